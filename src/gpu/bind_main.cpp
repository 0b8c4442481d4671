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
  m.def("device_attr", [](int dev) {
    hipDeviceProp_t p;
    if (hipGetDeviceProperties(&p, dev) != hipSuccess)
      throw std::runtime_error("hipGetDeviceProperties failed");
    py::dict d;
    d["name"] = std::string(p.name);
    d["gcn_arch_name"] = std::string(p.gcnArchName);
    d["is_integrated"] = p.integrated != 0;
    d["total_global_mem"] = (int64_t)p.totalGlobalMem;
    d["multi_processor_count"] = p.multiProcessorCount;
    d["shared_mem_per_block"] = (int64_t)p.sharedMemPerBlock;
    d["warp_size"] = p.warpSize;
    d["clock_rate_khz"] = p.clockRate;
    return d;
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
  register_lists(m);
  register_tools(m);
  register_dec128(m);
}
