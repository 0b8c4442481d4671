// Bindings for device shuffle serialization (Kudo GPU serializer).
#include "srj_bind.hpp"

extern "C" {
void srj_segmented_copy(const void*, const int64_t*, int32_t, int64_t,
                        hipStream_t);
void srj_validity_merge(const void*, const int64_t*, int32_t, int64_t,
                        hipStream_t);
void srj_offsets_rebase(const void*, const int64_t*, int32_t, int64_t,
                        hipStream_t);
void srj_gather_i32_at(const uint64_t*, int32_t, int32_t*, hipStream_t);
}

void register_shuffle(py::module_& m) {
  m.def("segmented_copy", [](uintptr_t segs, uintptr_t prefix, int32_t nsegs,
                             int64_t total_chunks, uintptr_t stream) {
    srj_segmented_copy(as_ptr<void>(segs), as_ptr<int64_t>(prefix), nsegs,
                       total_chunks, as_stream(stream));
    check_hip("segmented_copy");
  });
  m.def("validity_merge", [](uintptr_t segs, uintptr_t prefix, int32_t nsegs,
                             int64_t total_words, uintptr_t stream) {
    srj_validity_merge(as_ptr<void>(segs), as_ptr<int64_t>(prefix), nsegs,
                       total_words, as_stream(stream));
    check_hip("validity_merge");
  });
  m.def("offsets_rebase", [](uintptr_t segs, uintptr_t prefix, int32_t nsegs,
                             int64_t total, uintptr_t stream) {
    srj_offsets_rebase(as_ptr<void>(segs), as_ptr<int64_t>(prefix), nsegs, total,
                       as_stream(stream));
    check_hip("offsets_rebase");
  });
  m.def("gather_i32_at", [](uintptr_t addrs, int32_t n, uintptr_t out,
                            uintptr_t stream) {
    srj_gather_i32_at(as_ptr<uint64_t>(addrs), n, as_ptr<int32_t>(out),
                      as_stream(stream));
    check_hip("gather_i32_at");
  });
}
