// Shared pybind11 helpers for the _gpu extension. Bindings receive raw device
// pointers (Python ints from torch Tensor.data_ptr()) plus the current HIP
// stream; no torch headers are needed, keeping native compiles fast and the
// extension dependent only on libamdhip64.
#pragma once
#define __HIP_PLATFORM_AMD__ 1
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <hip/hip_runtime_api.h>
#include <cstdint>
#include <stdexcept>
#include <string>

namespace py = pybind11;

template <typename T>
inline T* as_ptr(uintptr_t p) { return reinterpret_cast<T*>(p); }

inline hipStream_t as_stream(uintptr_t s) { return reinterpret_cast<hipStream_t>(s); }

// Call after every kernel launch batch: surfaces launch errors as Python
// RuntimeError (the analog of the reference's CudaException JNI translation).
inline void check_hip(const char* what) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    throw std::runtime_error(std::string("HIP error in ") + what + ": " +
                             hipGetErrorString(e));
  }
}

void register_hash(py::module_& m);
void register_hashtable(py::module_& m);
void register_rowconv(py::module_& m);
void register_shuffle(py::module_& m);
void register_cast(py::module_& m);
void register_copying(py::module_& m);
void register_misc2(py::module_& m);
void register_json(py::module_& m);
void register_misc(py::module_& m);
void register_parquet(py::module_& m);
void register_sort(py::module_& m);
void register_lists(py::module_& m);
void register_tools(py::module_& m);
void register_dec128(py::module_& m);
