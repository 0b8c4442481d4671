// rocTX range markers (reference nvtx_ranges.hpp SRJ_FUNC_RANGE in the "srj"
// domain — here via rocprofiler-sdk-roctx, visible in rocprofv3 traces).
#include "srj_bind.hpp"

#include <rocprofiler-sdk-roctx/roctx.h>

extern "C" void srj_pb_decode(const void*, int64_t, void*, int32_t, uint8_t*,
                              int32_t, hipStream_t);
extern "C" void srj_install_ra_hooks(uintptr_t, uintptr_t, uintptr_t,
                                     uintptr_t);
extern "C" void srj_clear_ra_hooks();
extern "C" void srj_set_device_pool_limit(long long);
extern "C" long long srj_device_pool_used();

void register_tools(py::module_& m) {
  // torch pluggable-allocator bridge (src/gpu/torch_alloc.hip)
  m.def("install_ra_hooks", [](uintptr_t a, uintptr_t b, uintptr_t c,
                               uintptr_t d) {
    srj_install_ra_hooks(a, b, c, d);
  });
  m.def("clear_ra_hooks", [] { srj_clear_ra_hooks(); });
  m.def("set_device_pool_limit", [](long long b) {
    srj_set_device_pool_limit(b);
  });
  m.def("device_pool_used", [] { return srj_device_pool_used(); });
  m.def("roctx_range_push", [](const std::string& name) {
    return roctxRangePush(name.c_str());
  });
  m.def("roctx_range_pop", [] { return roctxRangePop(); });
  m.def("roctx_mark", [](const std::string& name) { roctxMarkA(name.c_str()); });
  m.def("roctx_range_start", [](const std::string& name) {
    return (int64_t)roctxRangeStartA(name.c_str());
  });
  m.def("roctx_range_stop", [](int64_t id) { roctxRangeStop((roctx_range_id_t)id); });
  m.def("pb_decode", [](uintptr_t in, int64_t n, uintptr_t fields,
                        int32_t nfields, uintptr_t row_ok, int32_t write_bytes,
                        uintptr_t stream) {
    srj_pb_decode(as_ptr<void>(in), n, as_ptr<void>(fields), nfields,
                  as_ptr<uint8_t>(row_ok), write_bytes, as_stream(stream));
    check_hip("pb_decode");
  });
}
