// Bindings for misc ops (case_when, bloom, zorder, hex, uuid, substring_index,
// literal_range, Aggregation64Utils, ANSI multiply, datetime rebase/trunc).
#include "srj_bind.hpp"

extern "C" {
void srj_select_first_true(const void*, const int32_t*, int32_t, int64_t,
                           int32_t*, hipStream_t);
void srj_bloom_filter(uint32_t*, int64_t, const int64_t*, const uint8_t*,
                      int64_t, int32_t, int32_t, int32_t, int32_t, uint8_t*,
                      uint8_t*, hipStream_t);
void srj_bitmask_or(const uint32_t*, uint32_t*, int64_t, hipStream_t);
void srj_interleave_bits(const void*, const int32_t*, int32_t, int32_t, int64_t,
                         uint8_t*, hipStream_t);
void srj_hilbert_index(const void*, const int32_t*, int32_t, int32_t, int64_t,
                       int64_t*, hipStream_t);
void srj_bytes_to_hex(const void*, int64_t, int32_t, int32_t*, const int32_t*,
                      char*, uint8_t*, hipStream_t);
void srj_uuid(int64_t, uint64_t, char*, hipStream_t);
void srj_substring_index(const void*, const char*, int32_t, int32_t, int64_t,
                         int32_t, int32_t*, const int32_t*, char*, uint8_t*,
                         hipStream_t);
void srj_literal_range(const void*, const char*, int32_t, int32_t,
                       uint32_t, uint32_t,
                       int64_t, uint8_t*, uint8_t*, hipStream_t);
void srj_extract_chunk32(const int64_t*, const uint8_t*, int64_t, int32_t,
                         int64_t*, hipStream_t);
void srj_combine_chunks(const int64_t*, const int64_t*, int64_t, int64_t*,
                        uint8_t*, hipStream_t);
void srj_multiply_i64(const int64_t*, const uint8_t*, const int64_t*,
                      const uint8_t*, int64_t, int64_t*, uint8_t*, int64_t*,
                      hipStream_t);
void srj_rebase_days(const int32_t*, const uint8_t*, int64_t, int32_t, int32_t*,
                     hipStream_t);
void srj_trunc_timestamp(const int64_t*, const uint8_t*, int64_t, int32_t,
                         int64_t*, hipStream_t);
}

void register_misc(py::module_& m) {
  m.def("select_first_true", [](uintptr_t cols, uintptr_t top, int32_t ncols,
                                int64_t n, uintptr_t out, uintptr_t stream) {
    srj_select_first_true(as_ptr<void>(cols), as_ptr<int32_t>(top), ncols, n,
                          as_ptr<int32_t>(out), as_stream(stream));
    check_hip("select_first_true");
  });
  m.def("bloom_filter", [](uintptr_t bits, int64_t fbits, uintptr_t input,
                           uintptr_t valid, int64_t n, int32_t num_hashes,
                           int32_t seed, int32_t version, int32_t probe,
                           uintptr_t out, uintptr_t out_valid, uintptr_t stream) {
    srj_bloom_filter(as_ptr<uint32_t>(bits), fbits, as_ptr<int64_t>(input),
                     as_ptr<uint8_t>(valid), n, num_hashes, seed, version, probe,
                     as_ptr<uint8_t>(out), as_ptr<uint8_t>(out_valid),
                     as_stream(stream));
    check_hip("bloom_filter");
  });
  m.def("bitmask_or", [](uintptr_t src, uintptr_t dst, int64_t nwords,
                         uintptr_t stream) {
    srj_bitmask_or(as_ptr<uint32_t>(src), as_ptr<uint32_t>(dst), nwords,
                   as_stream(stream));
    check_hip("bitmask_or");
  });
  m.def("interleave_bits", [](uintptr_t cols, uintptr_t top, int32_t ncols,
                              int32_t width, int64_t n, uintptr_t out,
                              uintptr_t stream) {
    srj_interleave_bits(as_ptr<void>(cols), as_ptr<int32_t>(top), ncols, width, n,
                        as_ptr<uint8_t>(out), as_stream(stream));
    check_hip("interleave_bits");
  });
  m.def("hilbert_index", [](uintptr_t cols, uintptr_t top, int32_t ncols,
                            int32_t nbits, int64_t n, uintptr_t out,
                            uintptr_t stream) {
    srj_hilbert_index(as_ptr<void>(cols), as_ptr<int32_t>(top), ncols, nbits, n,
                      as_ptr<int64_t>(out), as_stream(stream));
    check_hip("hilbert_index");
  });
  m.def("bytes_to_hex", [](uintptr_t in, int64_t n, int32_t phase, uintptr_t lens,
                           uintptr_t offsets, uintptr_t chars, uintptr_t valid,
                           uintptr_t stream) {
    srj_bytes_to_hex(as_ptr<void>(in), n, phase, as_ptr<int32_t>(lens),
                     as_ptr<int32_t>(offsets), as_ptr<char>(chars),
                     as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("bytes_to_hex");
  });
  m.def("uuid", [](int64_t n, uint64_t seed, uintptr_t chars, uintptr_t stream) {
    srj_uuid(n, seed, as_ptr<char>(chars), as_stream(stream));
    check_hip("uuid");
  });
  m.def("substring_index", [](uintptr_t in, uintptr_t delim, int32_t delim_len,
                              int32_t count, int64_t n, int32_t phase,
                              uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                              uintptr_t valid, uintptr_t stream) {
    srj_substring_index(as_ptr<void>(in), as_ptr<char>(delim), delim_len, count,
                        n, phase, as_ptr<int32_t>(lens), as_ptr<int32_t>(offsets),
                        as_ptr<char>(chars), as_ptr<uint8_t>(valid),
                        as_stream(stream));
    check_hip("substring_index");
  });
  m.def("literal_range", [](uintptr_t in, uintptr_t lit, int32_t lit_len,
                            int32_t range_len, int32_t rs, int32_t re, int64_t n,
                            uintptr_t out, uintptr_t valid, uintptr_t stream) {
    srj_literal_range(as_ptr<void>(in), as_ptr<char>(lit), lit_len, range_len,
                      (uint32_t)rs, (uint32_t)re, n, as_ptr<uint8_t>(out),
                      as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("literal_range");
  });
  m.def("extract_chunk32", [](uintptr_t in, uintptr_t valid, int64_t n,
                              int32_t chunk, uintptr_t out, uintptr_t stream) {
    srj_extract_chunk32(as_ptr<int64_t>(in), as_ptr<uint8_t>(valid), n, chunk,
                        as_ptr<int64_t>(out), as_stream(stream));
    check_hip("extract_chunk32");
  });
  m.def("combine_chunks", [](uintptr_t lo, uintptr_t hi, int64_t n, uintptr_t out,
                             uintptr_t overflow, uintptr_t stream) {
    srj_combine_chunks(as_ptr<int64_t>(lo), as_ptr<int64_t>(hi), n,
                       as_ptr<int64_t>(out), as_ptr<uint8_t>(overflow),
                       as_stream(stream));
    check_hip("combine_chunks");
  });
  m.def("multiply_i64", [](uintptr_t a, uintptr_t va, uintptr_t b, uintptr_t vb,
                           int64_t n, uintptr_t out, uintptr_t out_valid,
                           uintptr_t err, uintptr_t stream) {
    srj_multiply_i64(as_ptr<int64_t>(a), as_ptr<uint8_t>(va), as_ptr<int64_t>(b),
                     as_ptr<uint8_t>(vb), n, as_ptr<int64_t>(out),
                     as_ptr<uint8_t>(out_valid), as_ptr<int64_t>(err),
                     as_stream(stream));
    check_hip("multiply_i64");
  });
  m.def("rebase_days", [](uintptr_t in, uintptr_t valid, int64_t n,
                          int32_t to_julian, uintptr_t out, uintptr_t stream) {
    srj_rebase_days(as_ptr<int32_t>(in), as_ptr<uint8_t>(valid), n, to_julian,
                    as_ptr<int32_t>(out), as_stream(stream));
    check_hip("rebase_days");
  });
  m.def("trunc_timestamp", [](uintptr_t in, uintptr_t valid, int64_t n,
                              int32_t unit, uintptr_t out, uintptr_t stream) {
    srj_trunc_timestamp(as_ptr<int64_t>(in), as_ptr<uint8_t>(valid), n, unit,
                        as_ptr<int64_t>(out), as_stream(stream));
    check_hip("trunc_timestamp");
  });
}
