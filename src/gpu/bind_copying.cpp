// Bindings for gather / partition primitives.
#include "srj_bind.hpp"

extern "C" {
void srj_gather_fixed(const void*, const uint8_t*, const int64_t*, int64_t, void*,
                      uint8_t*, int32_t, hipStream_t);
void srj_gather_validity(const uint8_t*, const int64_t*, int64_t, uint8_t*,
                         hipStream_t);
void srj_gather_str_lengths(const int32_t*, const int64_t*, int64_t, int32_t*,
                            hipStream_t);
void srj_gather_str_chars(const char*, const int32_t*, const uint8_t*,
                          const int64_t*, const int32_t*, int64_t, char*, uint8_t*,
                          hipStream_t);
void srj_partition_hist(const int32_t*, int64_t, int32_t, int64_t*, hipStream_t);
void srj_partition_scatter(const int32_t*, int64_t, int32_t, uint64_t*, int64_t*,
                           hipStream_t);
void srj_pmod(const int32_t*, int64_t, int32_t, int32_t*, hipStream_t);
}

void register_copying(py::module_& m) {
  m.def("gather_fixed",
        [](uintptr_t in, uintptr_t in_valid, uintptr_t map, int64_t n, uintptr_t out,
           uintptr_t out_valid, int32_t elem_size, uintptr_t stream) {
          srj_gather_fixed(as_ptr<void>(in), as_ptr<uint8_t>(in_valid),
                           as_ptr<int64_t>(map), n, as_ptr<void>(out),
                           as_ptr<uint8_t>(out_valid), elem_size, as_stream(stream));
          check_hip("gather_fixed");
        });
  m.def("gather_validity",
        [](uintptr_t in_valid, uintptr_t map, int64_t n, uintptr_t out_valid,
           uintptr_t stream) {
          srj_gather_validity(as_ptr<uint8_t>(in_valid), as_ptr<int64_t>(map),
                              n, as_ptr<uint8_t>(out_valid), as_stream(stream));
          check_hip("gather_validity");
        });
  m.def("gather_str_lengths",
        [](uintptr_t offsets, uintptr_t map, int64_t n, uintptr_t lens,
           uintptr_t stream) {
          srj_gather_str_lengths(as_ptr<int32_t>(offsets), as_ptr<int64_t>(map), n,
                                 as_ptr<int32_t>(lens), as_stream(stream));
          check_hip("gather_str_lengths");
        });
  m.def("gather_str_chars",
        [](uintptr_t in_chars, uintptr_t in_offsets, uintptr_t in_valid,
           uintptr_t map, uintptr_t out_offsets, int64_t n, uintptr_t out_chars,
           uintptr_t out_valid, uintptr_t stream) {
          srj_gather_str_chars(as_ptr<char>(in_chars), as_ptr<int32_t>(in_offsets),
                               as_ptr<uint8_t>(in_valid), as_ptr<int64_t>(map),
                               as_ptr<int32_t>(out_offsets), n, as_ptr<char>(out_chars),
                               as_ptr<uint8_t>(out_valid), as_stream(stream));
          check_hip("gather_str_chars");
        });
  m.def("partition_hist",
        [](uintptr_t parts, int64_t n, int32_t nparts, uintptr_t hist,
           uintptr_t stream) {
          srj_partition_hist(as_ptr<int32_t>(parts), n, nparts, as_ptr<int64_t>(hist),
                             as_stream(stream));
          check_hip("partition_hist");
        });
  m.def("partition_scatter",
        [](uintptr_t parts, int64_t n, int32_t nparts, uintptr_t cursors,
           uintptr_t perm, uintptr_t stream) {
          srj_partition_scatter(as_ptr<int32_t>(parts), n, nparts,
                                as_ptr<uint64_t>(cursors), as_ptr<int64_t>(perm),
                                as_stream(stream));
          check_hip("partition_scatter");
        });
  m.def("pmod", [](uintptr_t hash, int64_t n, int32_t nparts, uintptr_t out,
                   uintptr_t stream) {
    srj_pmod(as_ptr<int32_t>(hash), n, nparts, as_ptr<int32_t>(out), as_stream(stream));
    check_hip("pmod");
  });
}
