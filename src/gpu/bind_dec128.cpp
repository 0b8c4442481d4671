// Bindings for DECIMAL128 arithmetic (Java API parity: DecimalUtils.java).
#include "srj_bind.hpp"

extern "C" {
void srj_dec128_mul(const void*, const uint8_t*, const void*, const uint8_t*,
                    int64_t, int32_t, int32_t, int32_t, void*, uint8_t*,
                    int64_t*, hipStream_t);
void srj_dec128_div(const void*, const uint8_t*, const void*, const uint8_t*,
                    int64_t, int32_t, int32_t, int32_t, int32_t, int32_t,
                    int32_t, void*, uint8_t*, int64_t*, hipStream_t);
void srj_dec128_addsub(const void*, const uint8_t*, const void*, const uint8_t*,
                       int64_t, int32_t, int32_t, int32_t, int32_t, void*,
                       uint8_t*, int64_t*, hipStream_t);
}

void register_dec128(py::module_& m) {
  m.def("dec128_mul", [](uintptr_t a, uintptr_t va, uintptr_t b, uintptr_t vb,
                         int64_t n, int32_t scale_sum, int32_t out_scale,
                         int32_t out_precision, uintptr_t out, uintptr_t ov,
                         uintptr_t err, uintptr_t stream) {
    srj_dec128_mul(as_ptr<void>(a), as_ptr<uint8_t>(va), as_ptr<void>(b),
                   as_ptr<uint8_t>(vb), n, scale_sum, out_scale, out_precision,
                   as_ptr<void>(out), as_ptr<uint8_t>(ov), as_ptr<int64_t>(err),
                   as_stream(stream));
    check_hip("dec128_mul");
  });
  m.def("dec128_div", [](uintptr_t a, uintptr_t va, uintptr_t b, uintptr_t vb,
                         int64_t n, int32_t s1, int32_t s2, int32_t out_scale,
                         int32_t out_precision, int32_t integer_div,
                         int32_t remainder, uintptr_t out, uintptr_t ov,
                         uintptr_t err, uintptr_t stream) {
    srj_dec128_div(as_ptr<void>(a), as_ptr<uint8_t>(va), as_ptr<void>(b),
                   as_ptr<uint8_t>(vb), n, s1, s2, out_scale, out_precision,
                   integer_div, remainder, as_ptr<void>(out), as_ptr<uint8_t>(ov),
                   as_ptr<int64_t>(err), as_stream(stream));
    check_hip("dec128_div");
  });
  m.def("dec128_addsub", [](uintptr_t a, uintptr_t va, uintptr_t b, uintptr_t vb,
                            int64_t n, int32_t up_a, int32_t up_b, int32_t sub,
                            int32_t out_precision, uintptr_t out, uintptr_t ov,
                            uintptr_t err, uintptr_t stream) {
    srj_dec128_addsub(as_ptr<void>(a), as_ptr<uint8_t>(va), as_ptr<void>(b),
                      as_ptr<uint8_t>(vb), n, up_a, up_b, sub, out_precision,
                      as_ptr<void>(out), as_ptr<uint8_t>(ov),
                      as_ptr<int64_t>(err), as_stream(stream));
    check_hip("dec128_addsub");
  });
}
