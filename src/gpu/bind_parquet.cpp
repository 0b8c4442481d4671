// Bindings for parquet page decode.
#include "srj_bind.hpp"

extern "C" {
void srj_rle_decode(const void*, int32_t, hipStream_t);
void srj_scatter_fixed(const void*, int32_t, uint8_t*, hipStream_t);
void srj_string_plain_index(const void*, int32_t, hipStream_t);
void srj_string_copy(const void*, int32_t, int32_t, int32_t*, const int32_t*,
                     uint8_t*, hipStream_t);
void srj_def_to_validity(const uint8_t*, int64_t, uint8_t*, hipStream_t);
void srj_pq_snappy_decomp(const void*, int32_t, hipStream_t);
void srj_gather_u8_at(const uint64_t*, int64_t, uint8_t*, hipStream_t);
void srj_pq_flba_dec128(const void*, int32_t, int64_t*, hipStream_t);
void srj_pq_delta_binpack(const void*, int32_t, hipStream_t);
void srj_pq_str_off(const void*, int32_t, hipStream_t);
void srj_pq_dba_reconstruct(const void*, int32_t, hipStream_t);
void srj_pq_bss(const void*, int32_t, hipStream_t);
}

void register_parquet(py::module_& m) {
  m.def("pq_delta_binpack", [](uintptr_t descs, int32_t n, uintptr_t stream) {
    srj_pq_delta_binpack(as_ptr<void>(descs), n, as_stream(stream));
    check_hip("pq_delta_binpack");
  });
  m.def("pq_str_off", [](uintptr_t descs, int32_t n, uintptr_t stream) {
    srj_pq_str_off(as_ptr<void>(descs), n, as_stream(stream));
    check_hip("pq_str_off");
  });
  m.def("pq_dba_reconstruct", [](uintptr_t descs, int32_t n,
                                 uintptr_t stream) {
    srj_pq_dba_reconstruct(as_ptr<void>(descs), n, as_stream(stream));
    check_hip("pq_dba_reconstruct");
  });
  m.def("pq_bss", [](uintptr_t descs, int32_t n, uintptr_t stream) {
    srj_pq_bss(as_ptr<void>(descs), n, as_stream(stream));
    check_hip("pq_bss");
  });
  m.def("pq_snappy_decomp", [](uintptr_t descs, int32_t n, uintptr_t stream) {
    srj_pq_snappy_decomp(as_ptr<void>(descs), n, as_stream(stream));
    check_hip("pq_snappy_decomp");
  });
  m.def("gather_u8_at", [](uintptr_t addrs, int64_t n, uintptr_t out,
                           uintptr_t stream) {
    srj_gather_u8_at(as_ptr<uint64_t>(addrs), n, as_ptr<uint8_t>(out),
                     as_stream(stream));
    check_hip("gather_u8_at");
  });
  m.def("pq_flba_dec128", [](uintptr_t descs, int32_t npages, uintptr_t out,
                             uintptr_t stream) {
    srj_pq_flba_dec128(as_ptr<void>(descs), npages, as_ptr<int64_t>(out),
                       as_stream(stream));
    check_hip("pq_flba_dec128");
  });
  m.def("pq_rle_decode", [](uintptr_t descs, int32_t npages, uintptr_t stream) {
    srj_rle_decode(as_ptr<void>(descs), npages, as_stream(stream));
    check_hip("pq_rle_decode");
  });
  m.def("pq_scatter_fixed", [](uintptr_t descs, int32_t npages, uintptr_t out,
                               uintptr_t stream) {
    srj_scatter_fixed(as_ptr<void>(descs), npages, as_ptr<uint8_t>(out),
                      as_stream(stream));
    check_hip("pq_scatter_fixed");
  });
  m.def("pq_string_plain_index", [](uintptr_t descs, int32_t npages,
                                    uintptr_t stream) {
    srj_string_plain_index(as_ptr<void>(descs), npages, as_stream(stream));
    check_hip("pq_string_plain_index");
  });
  m.def("pq_string_copy", [](uintptr_t descs, int32_t npages, int32_t phase,
                             uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                             uintptr_t stream) {
    srj_string_copy(as_ptr<void>(descs), npages, phase, as_ptr<int32_t>(lens),
                    as_ptr<int32_t>(offsets), as_ptr<uint8_t>(chars),
                    as_stream(stream));
    check_hip("pq_string_copy");
  });
  m.def("pq_def_to_validity", [](uintptr_t def, int64_t nrows, uintptr_t validity,
                                 uintptr_t stream) {
    srj_def_to_validity(as_ptr<uint8_t>(def), nrows, as_ptr<uint8_t>(validity),
                        as_stream(stream));
    check_hip("pq_def_to_validity");
  });
}
