// (no pending stubs)
#include "srj_bind.hpp"
