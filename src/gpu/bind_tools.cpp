// rocTX range markers (reference nvtx_ranges.hpp SRJ_FUNC_RANGE in the "srj"
// domain — here via rocprofiler-sdk-roctx, visible in rocprofv3 traces).
#include "srj_bind.hpp"

#include <rocprofiler-sdk-roctx/roctx.h>

void register_tools(py::module_& m) {
  m.def("roctx_range_push", [](const std::string& name) {
    return roctxRangePush(name.c_str());
  });
  m.def("roctx_range_pop", [] { return roctxRangePop(); });
  m.def("roctx_mark", [](const std::string& name) { roctxMarkA(name.c_str()); });
  m.def("roctx_range_start", [](const std::string& name) {
    return (int64_t)roctxRangeStartA(name.c_str());
  });
  m.def("roctx_range_stop", [](int64_t id) { roctxRangeStop((roctx_range_id_t)id); });
}
