// Bindings for JCUDF row conversion (Java API parity: RowConversion.java).
#include "srj_bind.hpp"

extern "C" {
void srj_to_rows(const void*, int32_t, int64_t, int32_t, int32_t, uint8_t*,
                 hipStream_t);
void srj_from_rows(const void*, int32_t, int64_t, int32_t, int32_t,
                   const uint8_t*, hipStream_t);
void srj_var_row_sizes(const void*, int32_t, int64_t, int32_t, int32_t*,
                       hipStream_t);
void srj_to_rows_var(const void*, const uint64_t*, int32_t, int64_t, int32_t,
                     int32_t, const int32_t*, uint8_t*, hipStream_t);
void srj_from_rows_var(const void*, const uint64_t*, const uint64_t*, int32_t,
                       int64_t, int32_t, const int32_t*, const uint8_t*,
                       int32_t, hipStream_t);
}

void register_rowconv(py::module_& m) {
  m.def("to_rows", [](uintptr_t cols, int32_t ncols, int64_t nrows,
                      int32_t row_size, int32_t validity_off, uintptr_t out,
                      uintptr_t stream) {
    srj_to_rows(as_ptr<void>(cols), ncols, nrows, row_size, validity_off,
                as_ptr<uint8_t>(out), as_stream(stream));
    check_hip("to_rows");
  });
  m.def("var_row_sizes", [](uintptr_t cols, int32_t ncols, int64_t n,
                            int32_t fixed_size, uintptr_t sizes,
                            uintptr_t stream) {
    srj_var_row_sizes(as_ptr<void>(cols), ncols, n, fixed_size,
                      as_ptr<int32_t>(sizes), as_stream(stream));
    check_hip("var_row_sizes");
  });
  m.def("to_rows_var", [](uintptr_t cols, uintptr_t char_ptrs, int32_t ncols,
                          int64_t n, int32_t fixed_size, int32_t validity_off,
                          uintptr_t row_offsets, uintptr_t out,
                          uintptr_t stream) {
    srj_to_rows_var(as_ptr<void>(cols), as_ptr<uint64_t>(char_ptrs), ncols, n,
                    fixed_size, validity_off, as_ptr<int32_t>(row_offsets),
                    as_ptr<uint8_t>(out), as_stream(stream));
    check_hip("to_rows_var");
  });
  m.def("from_rows_var", [](uintptr_t cols, uintptr_t char_ptrs,
                            uintptr_t len_ptrs, int32_t ncols, int64_t n,
                            int32_t validity_off, uintptr_t row_offsets,
                            uintptr_t in, int32_t phase, uintptr_t stream) {
    srj_from_rows_var(as_ptr<void>(cols), as_ptr<uint64_t>(char_ptrs),
                      as_ptr<uint64_t>(len_ptrs), ncols, n, validity_off,
                      as_ptr<int32_t>(row_offsets), as_ptr<uint8_t>(in), phase,
                      as_stream(stream));
    check_hip("from_rows_var");
  });
  m.def("from_rows", [](uintptr_t cols, int32_t ncols, int64_t nrows,
                        int32_t row_size, int32_t validity_off, uintptr_t in,
                        uintptr_t stream) {
    srj_from_rows(as_ptr<void>(cols), ncols, nrows, row_size, validity_off,
                  as_ptr<uint8_t>(in), as_stream(stream));
    check_hip("from_rows");
  });
}
