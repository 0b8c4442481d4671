// _gpu extension entry point. Subsystem registration lives in bind_*.cpp so a
// new op family only touches its own translation unit.
#include "srj_bind.hpp"

PYBIND11_MODULE(_gpu, m) {
  m.doc() = "MI355X-native (gfx950) HIP kernels for spark_rapids_jni_amd";
  m.def("device_count", [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return e == hipSuccess ? n : 0;
  });
  m.def("synchronize", [] {
    if (hipDeviceSynchronize() != hipSuccess)
      throw std::runtime_error("hipDeviceSynchronize failed");
  });
  register_hash(m);
  register_hashtable(m);
  register_copying(m);
  register_sort(m);
  register_rowconv(m);
  register_shuffle(m);
  register_cast(m);
  register_misc2(m);
  register_json(m);
  register_misc(m);
  register_parquet(m);
}
