// Bindings for radix sort + sort-merge join.
#include "srj_bind.hpp"

extern "C" {
void srj_radix_hist(const uint64_t*, int64_t, int32_t, int64_t, int64_t*,
                    hipStream_t);
void srj_radix_scatter(const uint64_t*, const int64_t*, int64_t, int32_t,
                       int64_t, const int64_t*, uint64_t*, int64_t*,
                       hipStream_t);
void srj_bias_i64(const int64_t*, int64_t, uint64_t*, hipStream_t);
void srj_unbias_i64(const uint64_t*, int64_t, int64_t*, hipStream_t);
void srj_merge_join(const int64_t*, const int64_t*, int64_t, const int64_t*,
                    const uint8_t*, int64_t, uint64_t*, int32_t*, int64_t*,
                    int64_t, int32_t, hipStream_t);
}

void register_sort(py::module_& m) {
  m.def("radix_hist", [](uintptr_t keys, int64_t n, int32_t shift,
                         int64_t nblocks, uintptr_t hist, uintptr_t stream) {
    srj_radix_hist(as_ptr<uint64_t>(keys), n, shift, nblocks,
                   as_ptr<int64_t>(hist), as_stream(stream));
    check_hip("radix_hist");
  });
  m.def("radix_scatter", [](uintptr_t keys, uintptr_t payload, int64_t n,
                            int32_t shift, int64_t nblocks, uintptr_t offsets,
                            uintptr_t out_keys, uintptr_t out_payload,
                            uintptr_t stream) {
    srj_radix_scatter(as_ptr<uint64_t>(keys), as_ptr<int64_t>(payload), n, shift,
                      nblocks, as_ptr<int64_t>(offsets), as_ptr<uint64_t>(out_keys),
                      as_ptr<int64_t>(out_payload), as_stream(stream));
    check_hip("radix_scatter");
  });
  m.def("bias_i64", [](uintptr_t in, int64_t n, uintptr_t out, uintptr_t stream) {
    srj_bias_i64(as_ptr<int64_t>(in), n, as_ptr<uint64_t>(out), as_stream(stream));
    check_hip("bias_i64");
  });
  m.def("unbias_i64", [](uintptr_t in, int64_t n, uintptr_t out,
                         uintptr_t stream) {
    srj_unbias_i64(as_ptr<uint64_t>(in), n, as_ptr<int64_t>(out),
                   as_stream(stream));
    check_hip("unbias_i64");
  });
  m.def("merge_join", [](uintptr_t build_sorted, uintptr_t build_rows,
                         int64_t nbuild, uintptr_t probe, uintptr_t pvalid,
                         int64_t nprobe, uintptr_t counter, uintptr_t out_build,
                         uintptr_t out_probe, int64_t out_capacity, int32_t fill,
                         uintptr_t stream) {
    srj_merge_join(as_ptr<int64_t>(build_sorted), as_ptr<int64_t>(build_rows),
                   nbuild, as_ptr<int64_t>(probe), as_ptr<uint8_t>(pvalid), nprobe,
                   as_ptr<uint64_t>(counter), as_ptr<int32_t>(out_build),
                   as_ptr<int64_t>(out_probe), out_capacity, fill,
                   as_stream(stream));
    check_hip("merge_join");
  });
}
