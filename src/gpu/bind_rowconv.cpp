// Bindings for JCUDF row conversion (Java API parity: RowConversion.java).
#include "srj_bind.hpp"

extern "C" {
void srj_to_rows(const void*, int32_t, int64_t, int32_t, int32_t, uint8_t*,
                 hipStream_t);
void srj_from_rows(const void*, int32_t, int64_t, int32_t, int32_t,
                   const uint8_t*, hipStream_t);
}

void register_rowconv(py::module_& m) {
  m.def("to_rows", [](uintptr_t cols, int32_t ncols, int64_t nrows,
                      int32_t row_size, int32_t validity_off, uintptr_t out,
                      uintptr_t stream) {
    srj_to_rows(as_ptr<void>(cols), ncols, nrows, row_size, validity_off,
                as_ptr<uint8_t>(out), as_stream(stream));
    check_hip("to_rows");
  });
  m.def("from_rows", [](uintptr_t cols, int32_t ncols, int64_t nrows,
                        int32_t row_size, int32_t validity_off, uintptr_t in,
                        uintptr_t stream) {
    srj_from_rows(as_ptr<void>(cols), ncols, nrows, row_size, validity_off,
                  as_ptr<uint8_t>(in), as_stream(stream));
    check_hip("from_rows");
  });
}
