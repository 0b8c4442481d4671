// Temporary empty registrations for subsystems not yet implemented; each is
// replaced by its own bind_*.cpp as the corresponding kernels land.
#include "srj_bind.hpp"

void register_datetime(py::module_&) {}
