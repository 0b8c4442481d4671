// Bindings for join/aggregate hash tables (Java API parity: JoinPrimitives.java).
#include "srj_bind.hpp"

extern "C" {
void srj_join_build(const void*, const int32_t*, int32_t, int64_t, uint64_t*,
                    int64_t, hipStream_t);
void srj_join_probe_count(const void*, const int32_t*, const void*, const int32_t*,
                          int32_t, int64_t, const uint64_t*, int64_t, uint64_t*,
                          hipStream_t);
void srj_join_probe_fill(const void*, const int32_t*, const void*, const int32_t*,
                         int32_t, int64_t, const uint64_t*, int64_t, uint64_t*,
                         int32_t*, int64_t*, int64_t, uint8_t*, hipStream_t);
void srj_join_semi(const void*, const int32_t*, const void*, const int32_t*,
                   int32_t, int64_t, const uint64_t*, int64_t, uint64_t*,
                   int64_t*, int64_t, int32_t, hipStream_t);
void srj_groupby(const void*, const int32_t*, int32_t, int64_t, uint64_t*, int64_t,
                 const void*, int32_t, int32_t*, hipStream_t);
void srj_groupby_compact(const uint64_t*, int64_t, const void*, int32_t, uint64_t*,
                         int64_t*, int64_t*, int64_t, hipStream_t);
void srj_join_build_i64(const long long*, const uint8_t*, int64_t, void*, int64_t,
                        hipStream_t);
void srj_groupby_i64(const long long*, int64_t, void*, int64_t, const void*,
                     int32_t, int32_t*, hipStream_t);
void srj_groupby_i64_lds(const long long*, int64_t, void*, int64_t,
                         const void*, int32_t, const int64_t*, int32_t*,
                         hipStream_t);
void srj_groupby_compact_i64_count(const void*, int64_t, int64_t*,
                                   hipStream_t);
void srj_groupby_compact_i64_fill(const void*, int64_t, const void*, int32_t,
                                  const int64_t*, int64_t*, int64_t*, int64_t,
                                  hipStream_t);
void srj_groupby_compact_i64(const void*, int64_t, const void*, int32_t,
                             uint64_t*, int64_t*, int64_t*, int64_t,
                             hipStream_t);
void srj_part_hist(const long long*, int64_t, int32_t, int64_t*, hipStream_t);
void srj_part_scatter(const long long*, int64_t, int32_t, int64_t*, long long*,
                      int32_t*, hipStream_t);
void srj_join_probe_i64(const long long*, const uint8_t*, int64_t, const void*,
                        int64_t, uint64_t*, int32_t*, int64_t*, int64_t, uint8_t*,
                        int32_t, const int32_t*, hipStream_t);
}

void register_hashtable(py::module_& m) {
  m.def("join_build",
        [](uintptr_t cols, uintptr_t top, int32_t ntop, int64_t nrows,
           uintptr_t slots, int64_t capacity, uintptr_t stream) {
          srj_join_build(as_ptr<void>(cols), as_ptr<int32_t>(top), ntop, nrows,
                         as_ptr<uint64_t>(slots), capacity, as_stream(stream));
          check_hip("join_build");
        });
  m.def("join_probe_count",
        [](uintptr_t bcols, uintptr_t btop, uintptr_t pcols, uintptr_t ptop,
           int32_t ntop, int64_t nprobe, uintptr_t slots, int64_t capacity,
           uintptr_t counter, uintptr_t stream) {
          srj_join_probe_count(as_ptr<void>(bcols), as_ptr<int32_t>(btop),
                               as_ptr<void>(pcols), as_ptr<int32_t>(ptop), ntop,
                               nprobe, as_ptr<uint64_t>(slots), capacity,
                               as_ptr<uint64_t>(counter), as_stream(stream));
          check_hip("join_probe_count");
        });
  m.def("join_probe_fill",
        [](uintptr_t bcols, uintptr_t btop, uintptr_t pcols, uintptr_t ptop,
           int32_t ntop, int64_t nprobe, uintptr_t slots, int64_t capacity,
           uintptr_t counter, uintptr_t out_build, uintptr_t out_probe,
           int64_t out_capacity, uintptr_t build_matched, uintptr_t stream) {
          srj_join_probe_fill(as_ptr<void>(bcols), as_ptr<int32_t>(btop),
                              as_ptr<void>(pcols), as_ptr<int32_t>(ptop), ntop,
                              nprobe, as_ptr<uint64_t>(slots), capacity,
                              as_ptr<uint64_t>(counter), as_ptr<int32_t>(out_build),
                              as_ptr<int64_t>(out_probe), out_capacity,
                              as_ptr<uint8_t>(build_matched), as_stream(stream));
          check_hip("join_probe_fill");
        });
  m.def("join_semi",
        [](uintptr_t bcols, uintptr_t btop, uintptr_t pcols, uintptr_t ptop,
           int32_t ntop, int64_t nprobe, uintptr_t slots, int64_t capacity,
           uintptr_t counter, uintptr_t out_probe, int64_t out_capacity,
           int32_t anti, uintptr_t stream) {
          srj_join_semi(as_ptr<void>(bcols), as_ptr<int32_t>(btop),
                        as_ptr<void>(pcols), as_ptr<int32_t>(ptop), ntop, nprobe,
                        as_ptr<uint64_t>(slots), capacity, as_ptr<uint64_t>(counter),
                        as_ptr<int64_t>(out_probe), out_capacity, anti,
                        as_stream(stream));
          check_hip("join_semi");
        });
  m.def("join_build_i64",
        [](uintptr_t keys, uintptr_t valid, int64_t nrows, uintptr_t slots,
           int64_t capacity, uintptr_t stream) {
          srj_join_build_i64(as_ptr<long long>(keys), as_ptr<uint8_t>(valid), nrows,
                             as_ptr<void>(slots), capacity, as_stream(stream));
          check_hip("join_build_i64");
        });
  m.def("groupby_i64",
        [](uintptr_t keys, int64_t nrows, uintptr_t slots, int64_t capacity,
           uintptr_t aggs, int32_t naggs, uintptr_t overflow,
           uintptr_t stream) {
          srj_groupby_i64(as_ptr<long long>(keys), nrows, as_ptr<void>(slots),
                          capacity, as_ptr<void>(aggs), naggs,
                          as_ptr<int32_t>(overflow), as_stream(stream));
          check_hip("groupby_i64");
        });
  m.def("groupby_i64_lds",
        [](uintptr_t keys, int64_t nrows, uintptr_t slots, int64_t capacity,
           uintptr_t aggs, int32_t naggs, uintptr_t identities,
           uintptr_t overflow, uintptr_t stream) {
          srj_groupby_i64_lds(as_ptr<long long>(keys), nrows,
                              as_ptr<void>(slots), capacity,
                              as_ptr<void>(aggs), naggs,
                              as_ptr<int64_t>(identities),
                              as_ptr<int32_t>(overflow), as_stream(stream));
          check_hip("groupby_i64_lds");
        });
  m.def("groupby_compact_i64",
        [](uintptr_t slots, int64_t capacity1, uintptr_t aggs, int32_t naggs,
           uintptr_t counter, uintptr_t out_repr, uintptr_t out_agg,
           int64_t out_capacity, uintptr_t stream) {
          srj_groupby_compact_i64(as_ptr<void>(slots), capacity1,
                                  as_ptr<void>(aggs), naggs,
                                  as_ptr<uint64_t>(counter),
                                  as_ptr<int64_t>(out_repr),
                                  as_ptr<int64_t>(out_agg), out_capacity,
                                  as_stream(stream));
          check_hip("groupby_compact_i64");
        });
  m.def("groupby_compact_i64_count",
        [](uintptr_t slots, int64_t capacity1, uintptr_t blk_counts,
           uintptr_t stream) {
          srj_groupby_compact_i64_count(as_ptr<void>(slots), capacity1,
                                        as_ptr<int64_t>(blk_counts),
                                        as_stream(stream));
          check_hip("groupby_compact_i64_count");
        });
  m.def("groupby_compact_i64_fill",
        [](uintptr_t slots, int64_t capacity1, uintptr_t aggs, int32_t naggs,
           uintptr_t blk_bases, uintptr_t out_repr, uintptr_t out_agg,
           int64_t out_capacity, uintptr_t stream) {
          srj_groupby_compact_i64_fill(as_ptr<void>(slots), capacity1,
                                       as_ptr<void>(aggs), naggs,
                                       as_ptr<int64_t>(blk_bases),
                                       as_ptr<int64_t>(out_repr),
                                       as_ptr<int64_t>(out_agg), out_capacity,
                                       as_stream(stream));
          check_hip("groupby_compact_i64_fill");
        });
  m.def("join_probe_i64",
        [](uintptr_t probe, uintptr_t pvalid, int64_t nprobe, uintptr_t slots,
           int64_t capacity, uintptr_t counter, uintptr_t out_build,
           uintptr_t out_probe, int64_t out_capacity, uintptr_t build_matched,
           int32_t fill, uintptr_t idxmap, uintptr_t stream) {
          srj_join_probe_i64(as_ptr<long long>(probe), as_ptr<uint8_t>(pvalid),
                             nprobe, as_ptr<void>(slots), capacity,
                             as_ptr<uint64_t>(counter), as_ptr<int32_t>(out_build),
                             as_ptr<int64_t>(out_probe), out_capacity,
                             as_ptr<uint8_t>(build_matched), fill,
                             as_ptr<int32_t>(idxmap), as_stream(stream));
          check_hip("join_probe_i64");
        });
  m.def("part_hist",
        [](uintptr_t keys, int64_t n, int32_t pbits, uintptr_t hist,
           uintptr_t stream) {
          srj_part_hist(as_ptr<long long>(keys), n, pbits,
                        as_ptr<int64_t>(hist), as_stream(stream));
          check_hip("part_hist");
        });
  m.def("part_scatter",
        [](uintptr_t keys, int64_t n, int32_t pbits, uintptr_t cursors,
           uintptr_t out_keys, uintptr_t out_idx, uintptr_t stream) {
          srj_part_scatter(as_ptr<long long>(keys), n, pbits,
                           as_ptr<int64_t>(cursors), as_ptr<long long>(out_keys),
                           as_ptr<int32_t>(out_idx), as_stream(stream));
          check_hip("part_scatter");
        });
  m.def("groupby",
        [](uintptr_t cols, uintptr_t top, int32_t ntop, int64_t nrows,
           uintptr_t slots, int64_t capacity, uintptr_t aggs, int32_t naggs,
           uintptr_t overflow, uintptr_t stream) {
          srj_groupby(as_ptr<void>(cols), as_ptr<int32_t>(top), ntop, nrows,
                      as_ptr<uint64_t>(slots), capacity, as_ptr<void>(aggs), naggs,
                      as_ptr<int32_t>(overflow), as_stream(stream));
          check_hip("groupby");
        });
  m.def("groupby_compact",
        [](uintptr_t slots, int64_t capacity, uintptr_t aggs, int32_t naggs,
           uintptr_t counter, uintptr_t out_repr, uintptr_t out_agg,
           int64_t out_capacity, uintptr_t stream) {
          srj_groupby_compact(as_ptr<uint64_t>(slots), capacity, as_ptr<void>(aggs),
                              naggs, as_ptr<uint64_t>(counter),
                              as_ptr<int64_t>(out_repr), as_ptr<int64_t>(out_agg),
                              out_capacity, as_stream(stream));
          check_hip("groupby_compact");
        });
}
