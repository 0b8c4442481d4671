// Bindings for Spark-exact casts (Java API parity: CastStrings.java).
#include "srj_bind.hpp"

extern "C" {
void srj_string_to_int(const void*, int64_t, int32_t, int32_t, void*, uint8_t*,
                       int64_t*, hipStream_t);
void srj_string_to_bool(const void*, int64_t, int8_t*, uint8_t*, int64_t*,
                        hipStream_t);
void srj_string_to_float(const void*, int64_t, int32_t, void*, uint8_t*,
                         int64_t*, hipStream_t);
void srj_string_to_decimal(const void*, int64_t, int32_t, int32_t, int32_t,
                           void*, uint8_t*, int64_t*, hipStream_t);
void srj_string_to_date(const void*, int64_t, int64_t, int32_t*, uint8_t*,
                        int64_t*, hipStream_t);
void srj_string_to_timestamp(const void*, int64_t, int64_t, int64_t, int64_t,
                             int64_t*, uint8_t*, int64_t*, hipStream_t);
void srj_integer_to_string(const void*, int64_t, int32_t, int32_t*,
                           const int32_t*, char*, uint8_t*, hipStream_t);
void srj_format_number(const double*, const uint8_t*, int64_t, int32_t,
                       int32_t, int32_t*, const int32_t*, char*, uint8_t*,
                       hipStream_t);
void srj_float_to_string(const void*, const uint8_t*, int64_t, int32_t, int32_t,
                         int32_t*, const int32_t*, char*, uint8_t*, hipStream_t);
void srj_parse_timestamp_fmt(const void*, int64_t, const void*, int32_t,
                             int32_t, int64_t, int64_t*, uint8_t*, int64_t*,
                             hipStream_t);
}

void register_cast(py::module_& m) {
  m.def("string_to_int", [](uintptr_t in, int64_t n, int32_t strip, int32_t width,
                            uintptr_t out, uintptr_t valid, uintptr_t err,
                            uintptr_t stream) {
    srj_string_to_int(as_ptr<void>(in), n, strip, width, as_ptr<void>(out),
                      as_ptr<uint8_t>(valid), as_ptr<int64_t>(err),
                      as_stream(stream));
    check_hip("string_to_int");
  });
  m.def("string_to_bool", [](uintptr_t in, int64_t n, uintptr_t out,
                             uintptr_t valid, uintptr_t err, uintptr_t stream) {
    srj_string_to_bool(as_ptr<void>(in), n, as_ptr<int8_t>(out),
                       as_ptr<uint8_t>(valid), as_ptr<int64_t>(err),
                       as_stream(stream));
    check_hip("string_to_bool");
  });
  m.def("string_to_float", [](uintptr_t in, int64_t n, int32_t width,
                              uintptr_t out, uintptr_t valid, uintptr_t err,
                              uintptr_t stream) {
    srj_string_to_float(as_ptr<void>(in), n, width, as_ptr<void>(out),
                        as_ptr<uint8_t>(valid), as_ptr<int64_t>(err),
                        as_stream(stream));
    check_hip("string_to_float");
  });
  m.def("string_to_decimal", [](uintptr_t in, int64_t n, int32_t precision,
                                int32_t scale, int32_t width, uintptr_t out,
                                uintptr_t valid, uintptr_t err, uintptr_t stream) {
    srj_string_to_decimal(as_ptr<void>(in), n, precision, scale, width,
                          as_ptr<void>(out), as_ptr<uint8_t>(valid),
                          as_ptr<int64_t>(err), as_stream(stream));
    check_hip("string_to_decimal");
  });
  m.def("string_to_date", [](uintptr_t in, int64_t n, int64_t today,
                             uintptr_t out, uintptr_t valid, uintptr_t err,
                             uintptr_t stream) {
    srj_string_to_date(as_ptr<void>(in), n, today, as_ptr<int32_t>(out),
                       as_ptr<uint8_t>(valid), as_ptr<int64_t>(err),
                       as_stream(stream));
    check_hip("string_to_date");
  });
  m.def("string_to_timestamp", [](uintptr_t in, int64_t n, int64_t now_us,
                                  int64_t today, int64_t tz_off, uintptr_t out,
                                  uintptr_t valid, uintptr_t err,
                                  uintptr_t stream) {
    srj_string_to_timestamp(as_ptr<void>(in), n, now_us, today, tz_off,
                            as_ptr<int64_t>(out), as_ptr<uint8_t>(valid),
                            as_ptr<int64_t>(err), as_stream(stream));
    check_hip("string_to_timestamp");
  });
  m.def("parse_timestamp_fmt", [](uintptr_t in, int64_t n, uintptr_t toks,
                                  int32_t ntoks, int32_t trail_nondigit,
                                  int64_t tz_off, uintptr_t out,
                                  uintptr_t valid, uintptr_t err,
                                  uintptr_t stream) {
    srj_parse_timestamp_fmt(as_ptr<void>(in), n, as_ptr<void>(toks), ntoks,
                            trail_nondigit, tz_off, as_ptr<int64_t>(out),
                            as_ptr<uint8_t>(valid),
                            as_ptr<int64_t>(err), as_stream(stream));
    check_hip("parse_timestamp_fmt");
  });
  m.def("format_number", [](uintptr_t in, uintptr_t valid, int64_t n,
                            int32_t d, int32_t phase, uintptr_t lens,
                            uintptr_t offsets, uintptr_t chars,
                            uintptr_t out_valid, uintptr_t stream) {
    srj_format_number(as_ptr<double>(in), as_ptr<uint8_t>(valid), n, d, phase,
                      as_ptr<int32_t>(lens), as_ptr<int32_t>(offsets),
                      as_ptr<char>(chars), as_ptr<uint8_t>(out_valid),
                      as_stream(stream));
    check_hip("format_number");
  });
  m.def("float_to_string", [](uintptr_t in, uintptr_t valid, int64_t n,
                              int32_t width, int32_t phase, uintptr_t lens,
                              uintptr_t offsets, uintptr_t chars,
                              uintptr_t out_valid, uintptr_t stream) {
    srj_float_to_string(as_ptr<void>(in), as_ptr<uint8_t>(valid), n, width, phase,
                        as_ptr<int32_t>(lens), as_ptr<int32_t>(offsets),
                        as_ptr<char>(chars), as_ptr<uint8_t>(out_valid),
                        as_stream(stream));
    check_hip("float_to_string");
  });
  m.def("integer_to_string", [](uintptr_t in, int64_t n, int32_t phase,
                                uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                                uintptr_t valid, uintptr_t stream) {
    srj_integer_to_string(as_ptr<void>(in), n, phase, as_ptr<int32_t>(lens),
                          as_ptr<int32_t>(offsets), as_ptr<char>(chars),
                          as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("integer_to_string");
  });
}
