// _host extension: CPU-only native subsystems (no HIP dependency).
#include <pybind11/pybind11.h>

namespace py = pybind11;

void register_resource_adaptor(py::module_& m);
void register_thrift(py::module_& m);

PYBIND11_MODULE(_host, m) {
  m.doc() = "CPU-side native subsystems for spark_rapids_jni_amd "
            "(OOM state machine, kudo serializer, parquet footer)";
  register_resource_adaptor(m);
  register_thrift(m);
}
