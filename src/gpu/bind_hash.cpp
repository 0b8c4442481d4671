// Bindings for the Spark-exact hash family (murmur3 / xxhash64 / hive_hash).
// Java API parity: com.nvidia.spark.rapids.jni.Hash (reference Hash.java).
#include "srj_bind.hpp"

extern "C" {
struct ColDescABI;  // opaque; Python packs the 48-byte layout itself
void srj_murmur3(const void*, const int32_t*, int32_t, int64_t, int32_t, int32_t*,
                 hipStream_t);
void srj_xxhash64(const void*, const int32_t*, int32_t, int64_t, int64_t, int64_t*,
                  hipStream_t);
void srj_hive_hash(const void*, const int32_t*, int32_t, int64_t, int32_t*,
                   hipStream_t);
void srj_count_set_bits(const uint8_t*, int64_t, uint64_t*, hipStream_t);
}

void register_hash(py::module_& m) {
  m.def("murmur3",
        [](uintptr_t cols, uintptr_t top, int32_t ntop, int64_t nrows, int32_t seed,
           uintptr_t out, uintptr_t stream) {
          srj_murmur3(as_ptr<void>(cols), as_ptr<int32_t>(top), ntop, nrows, seed,
                      as_ptr<int32_t>(out), as_stream(stream));
          check_hip("murmur3");
        });
  m.def("xxhash64",
        [](uintptr_t cols, uintptr_t top, int32_t ntop, int64_t nrows, int64_t seed,
           uintptr_t out, uintptr_t stream) {
          srj_xxhash64(as_ptr<void>(cols), as_ptr<int32_t>(top), ntop, nrows, seed,
                       as_ptr<int64_t>(out), as_stream(stream));
          check_hip("xxhash64");
        });
  m.def("hive_hash",
        [](uintptr_t cols, uintptr_t top, int32_t ntop, int64_t nrows, uintptr_t out,
           uintptr_t stream) {
          srj_hive_hash(as_ptr<void>(cols), as_ptr<int32_t>(top), ntop, nrows,
                        as_ptr<int32_t>(out), as_stream(stream));
          check_hip("hive_hash");
        });
  m.def("count_set_bits", [](uintptr_t mask, int64_t n, uintptr_t out, uintptr_t stream) {
    srj_count_set_bits(as_ptr<uint8_t>(mask), n, as_ptr<uint64_t>(out), as_stream(stream));
    check_hip("count_set_bits");
  });
}
