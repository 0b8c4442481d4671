// Bindings for list/map ops, iceberg, round_double, AST filter.
#include "srj_bind.hpp"

extern "C" {
void srj_list_slice(const int32_t*, const uint8_t*, int64_t, const int32_t*,
                    int32_t, const int32_t*, int32_t, int32_t*, int32_t*,
                    uint8_t*, int64_t*, hipStream_t);
void srj_list_slice_gather(const int32_t*, const int32_t*, int64_t, int64_t*,
                           hipStream_t);
void srj_validate_map(const int32_t*, const uint8_t*, int64_t, const void*,
                      int32_t, uint8_t*, hipStream_t);
void srj_sort_map(const int32_t*, int64_t, const void*, int32_t, int64_t*,
                  hipStream_t);
void srj_map_zip(const int32_t*, const int32_t*, int64_t, const void*, int32_t,
                 int32_t, int32_t, int32_t*, const int32_t*, int64_t*, int64_t*,
                 int64_t*, hipStream_t);
void srj_iceberg_bucket_long(const int64_t*, const uint8_t*, int64_t, int32_t,
                             int32_t*, uint8_t*, hipStream_t);
void srj_iceberg_bucket_decimal(const void*, const uint8_t*, int64_t,
                                int32_t, int32_t, int32_t*, uint8_t*,
                                hipStream_t);
void srj_iceberg_bucket_string(const void*, int64_t, int32_t, int32_t*,
                               uint8_t*, hipStream_t);
void srj_iceberg_truncate_long(const int64_t*, const uint8_t*, int64_t, int64_t,
                               int64_t*, hipStream_t);
void srj_iceberg_datetime(const void*, const uint8_t*, int64_t, int32_t,
                          int32_t, int32_t*, hipStream_t);
void srj_round_double(const double*, const uint8_t*, int64_t, int32_t, int32_t,
                      double*, hipStream_t);
void srj_ast_filter_pairs(const void*, const void*, int32_t, const int32_t*,
                          const int64_t*, int64_t, uint64_t*, int32_t*,
                          int64_t*, int64_t, int32_t, hipStream_t);
void srj_matched_rows(const int32_t*, int64_t, uint8_t*, hipStream_t);
}

void register_lists(py::module_& m) {
  m.def("list_slice", [](uintptr_t offs, uintptr_t valid, int64_t n,
                         uintptr_t sc, int32_t ss, uintptr_t lc, int32_t ls,
                         uintptr_t out_lens, uintptr_t child_start,
                         uintptr_t out_valid, uintptr_t err, uintptr_t stream) {
    srj_list_slice(as_ptr<int32_t>(offs), as_ptr<uint8_t>(valid), n,
                   as_ptr<int32_t>(sc), ss, as_ptr<int32_t>(lc), ls,
                   as_ptr<int32_t>(out_lens), as_ptr<int32_t>(child_start),
                   as_ptr<uint8_t>(out_valid), as_ptr<int64_t>(err),
                   as_stream(stream));
    check_hip("list_slice");
  });
  m.def("list_slice_gather", [](uintptr_t cs, uintptr_t oo, int64_t n,
                                uintptr_t gmap, uintptr_t stream) {
    srj_list_slice_gather(as_ptr<int32_t>(cs), as_ptr<int32_t>(oo), n,
                          as_ptr<int64_t>(gmap), as_stream(stream));
    check_hip("list_slice_gather");
  });
  m.def("validate_map", [](uintptr_t offs, uintptr_t rv, int64_t n,
                           uintptr_t cols, int32_t key_col, uintptr_t out,
                           uintptr_t stream) {
    srj_validate_map(as_ptr<int32_t>(offs), as_ptr<uint8_t>(rv), n,
                     as_ptr<void>(cols), key_col, as_ptr<uint8_t>(out),
                     as_stream(stream));
    check_hip("validate_map");
  });
  m.def("sort_map", [](uintptr_t offs, int64_t n, uintptr_t cols,
                       int32_t key_col, uintptr_t perm, uintptr_t stream) {
    srj_sort_map(as_ptr<int32_t>(offs), n, as_ptr<void>(cols), key_col,
                 as_ptr<int64_t>(perm), as_stream(stream));
    check_hip("sort_map");
  });
  m.def("map_zip", [](uintptr_t o1, uintptr_t o2, int64_t n, uintptr_t cols,
                      int32_t k1, int32_t k2, int32_t phase, uintptr_t counts,
                      uintptr_t oo, uintptr_t kmap, uintptr_t v1, uintptr_t v2,
                      uintptr_t stream) {
    srj_map_zip(as_ptr<int32_t>(o1), as_ptr<int32_t>(o2), n, as_ptr<void>(cols),
                k1, k2, phase, as_ptr<int32_t>(counts), as_ptr<int32_t>(oo),
                as_ptr<int64_t>(kmap), as_ptr<int64_t>(v1), as_ptr<int64_t>(v2),
                as_stream(stream));
    check_hip("map_zip");
  });
  m.def("iceberg_bucket_decimal", [](uintptr_t in, uintptr_t valid,
                                     int64_t n, int32_t width, int32_t nb,
                                     uintptr_t out, uintptr_t ov,
                                     uintptr_t stream) {
    srj_iceberg_bucket_decimal(as_ptr<void>(in), as_ptr<uint8_t>(valid), n,
                               width, nb, as_ptr<int32_t>(out),
                               as_ptr<uint8_t>(ov), as_stream(stream));
    check_hip("iceberg_bucket_decimal");
  });
  m.def("iceberg_bucket_long", [](uintptr_t in, uintptr_t valid, int64_t n,
                                  int32_t nb, uintptr_t out, uintptr_t ov,
                                  uintptr_t stream) {
    srj_iceberg_bucket_long(as_ptr<int64_t>(in), as_ptr<uint8_t>(valid), n, nb,
                            as_ptr<int32_t>(out), as_ptr<uint8_t>(ov),
                            as_stream(stream));
    check_hip("iceberg_bucket_long");
  });
  m.def("iceberg_bucket_string", [](uintptr_t in, int64_t n, int32_t nb,
                                    uintptr_t out, uintptr_t ov,
                                    uintptr_t stream) {
    srj_iceberg_bucket_string(as_ptr<void>(in), n, nb, as_ptr<int32_t>(out),
                              as_ptr<uint8_t>(ov), as_stream(stream));
    check_hip("iceberg_bucket_string");
  });
  m.def("iceberg_truncate_long", [](uintptr_t in, uintptr_t valid, int64_t n,
                                    int64_t w, uintptr_t out, uintptr_t stream) {
    srj_iceberg_truncate_long(as_ptr<int64_t>(in), as_ptr<uint8_t>(valid), n, w,
                              as_ptr<int64_t>(out), as_stream(stream));
    check_hip("iceberg_truncate_long");
  });
  m.def("iceberg_datetime", [](uintptr_t in, uintptr_t valid, int64_t n,
                               int32_t fm, int32_t part, uintptr_t out,
                               uintptr_t stream) {
    srj_iceberg_datetime(as_ptr<void>(in), as_ptr<uint8_t>(valid), n, fm, part,
                         as_ptr<int32_t>(out), as_stream(stream));
    check_hip("iceberg_datetime");
  });
  m.def("round_double", [](uintptr_t in, uintptr_t valid, int64_t n,
                           int32_t scale, int32_t he, uintptr_t out,
                           uintptr_t stream) {
    srj_round_double(as_ptr<double>(in), as_ptr<uint8_t>(valid), n, scale, he,
                     as_ptr<double>(out), as_stream(stream));
    check_hip("round_double");
  });
  m.def("ast_filter_pairs", [](uintptr_t cols, uintptr_t prog, int32_t nprog,
                               uintptr_t lmap, uintptr_t rmap, int64_t n,
                               uintptr_t counter, uintptr_t ol, uintptr_t orr,
                               int64_t cap, int32_t fill, uintptr_t stream) {
    srj_ast_filter_pairs(as_ptr<void>(cols), as_ptr<void>(prog), nprog,
                         as_ptr<int32_t>(lmap), as_ptr<int64_t>(rmap), n,
                         as_ptr<uint64_t>(counter), as_ptr<int32_t>(ol),
                         as_ptr<int64_t>(orr), cap, fill, as_stream(stream));
    check_hip("ast_filter_pairs");
  });
  m.def("matched_rows", [](uintptr_t gmap, int64_t n, uintptr_t flags,
                           uintptr_t stream) {
    srj_matched_rows(as_ptr<int32_t>(gmap), n, as_ptr<uint8_t>(flags),
                     as_stream(stream));
    check_hip("matched_rows");
  });
}
