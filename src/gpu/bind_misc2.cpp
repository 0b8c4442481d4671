// Bindings for SHA-2, HLL++, percentile, conv, parse_uri, GBK decode.
#include "srj_bind.hpp"

extern "C" {
void srj_sha2(const void*, int64_t, int32_t, const int32_t*, char*, uint8_t*,
              hipStream_t);
void srj_hllpp_update(const int64_t*, const uint8_t*, int64_t, int32_t,
                      int32_t*, hipStream_t);
void srj_hllpp_merge(const int32_t*, int32_t*, int64_t, hipStream_t);
void srj_hllpp_pack(const int32_t*, int64_t, int64_t*, hipStream_t);
void srj_hllpp_unpack(const int64_t*, int64_t, int32_t*, hipStream_t);
void srj_percentile_from_histogram(const int32_t*, const double*, const int64_t*,
                                   int64_t, const double*, int32_t, double*,
                                   uint8_t*, hipStream_t);
void srj_conv(const void*, int64_t, int32_t, int32_t, int32_t, int32_t*,
              const int32_t*, char*, uint8_t*, hipStream_t);
void srj_parse_uri(const void*, int64_t, int32_t, const char*, int32_t,
                   const void*, int32_t,
                   int32_t*, const int32_t*, char*, uint8_t*, hipStream_t);
void srj_gbk_decode(const void*, int64_t, int32_t, int32_t, int32_t*,
                    const int32_t*, char*, uint8_t*, int64_t*, hipStream_t);
void srj_tz_convert(const int64_t*, const uint8_t*, int64_t, const int64_t*,
                    const int64_t*, const int64_t*, const int32_t*, int32_t,
                    int32_t, int64_t*, hipStream_t);
}

void register_misc2(py::module_& m) {
  m.def("sha2", [](uintptr_t in, int64_t n, int32_t mode, uintptr_t offsets,
                   uintptr_t chars, uintptr_t valid, uintptr_t stream) {
    srj_sha2(as_ptr<void>(in), n, mode, as_ptr<int32_t>(offsets),
             as_ptr<char>(chars), as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("sha2");
  });
  m.def("hllpp_update", [](uintptr_t hashes, uintptr_t valid, int64_t n,
                           int32_t precision, uintptr_t regs, uintptr_t stream) {
    srj_hllpp_update(as_ptr<int64_t>(hashes), as_ptr<uint8_t>(valid), n,
                     precision, as_ptr<int32_t>(regs), as_stream(stream));
    check_hip("hllpp_update");
  });
  m.def("hllpp_merge", [](uintptr_t src, uintptr_t dst, int64_t nregs,
                          uintptr_t stream) {
    srj_hllpp_merge(as_ptr<int32_t>(src), as_ptr<int32_t>(dst), nregs,
                    as_stream(stream));
    check_hip("hllpp_merge");
  });
  m.def("hllpp_pack", [](uintptr_t regs, int64_t nregs, uintptr_t longs,
                         uintptr_t stream) {
    srj_hllpp_pack(as_ptr<int32_t>(regs), nregs, as_ptr<int64_t>(longs),
                   as_stream(stream));
    check_hip("hllpp_pack");
  });
  m.def("hllpp_unpack", [](uintptr_t longs, int64_t nregs, uintptr_t regs,
                           uintptr_t stream) {
    srj_hllpp_unpack(as_ptr<int64_t>(longs), nregs, as_ptr<int32_t>(regs),
                     as_stream(stream));
    check_hip("hllpp_unpack");
  });
  m.def("percentile_from_histogram",
        [](uintptr_t offsets, uintptr_t values, uintptr_t freqs, int64_t n,
           uintptr_t pcts, int32_t npct, uintptr_t out, uintptr_t valid,
           uintptr_t stream) {
          srj_percentile_from_histogram(
              as_ptr<int32_t>(offsets), as_ptr<double>(values),
              as_ptr<int64_t>(freqs), n, as_ptr<double>(pcts), npct,
              as_ptr<double>(out), as_ptr<uint8_t>(valid), as_stream(stream));
          check_hip("percentile_from_histogram");
        });
  m.def("conv", [](uintptr_t in, int64_t n, int32_t fb, int32_t tb, int32_t phase,
                   uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                   uintptr_t valid, uintptr_t stream) {
    srj_conv(as_ptr<void>(in), n, fb, tb, phase, as_ptr<int32_t>(lens),
             as_ptr<int32_t>(offsets), as_ptr<char>(chars), as_ptr<uint8_t>(valid),
             as_stream(stream));
    check_hip("conv");
  });
  m.def("parse_uri", [](uintptr_t in, int64_t n, int32_t part, uintptr_t qkey,
                        int32_t qkey_len, uintptr_t qcol, int32_t phase,
                        uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                        uintptr_t valid, uintptr_t stream) {
    srj_parse_uri(as_ptr<void>(in), n, part, as_ptr<char>(qkey), qkey_len,
                  as_ptr<void>(qcol), phase,
                  as_ptr<int32_t>(lens), as_ptr<int32_t>(offsets),
                  as_ptr<char>(chars), as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("parse_uri");
  });
  m.def("tz_convert", [](uintptr_t in, uintptr_t valid, int64_t n,
                         uintptr_t utc_us, uintptr_t local_us, uintptr_t off,
                         uintptr_t zoffs, int32_t zidx, int32_t to_utc,
                         uintptr_t out, uintptr_t stream) {
    srj_tz_convert(as_ptr<int64_t>(in), as_ptr<uint8_t>(valid), n,
                   as_ptr<int64_t>(utc_us), as_ptr<int64_t>(local_us),
                   as_ptr<int64_t>(off), as_ptr<int32_t>(zoffs), zidx, to_utc,
                   as_ptr<int64_t>(out), as_stream(stream));
    check_hip("tz_convert");
  });
  m.def("gbk_decode", [](uintptr_t in, int64_t n, int32_t report, int32_t phase,
                         uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                         uintptr_t valid, uintptr_t err, uintptr_t stream) {
    srj_gbk_decode(as_ptr<void>(in), n, report, phase, as_ptr<int32_t>(lens),
                   as_ptr<int32_t>(offsets), as_ptr<char>(chars),
                   as_ptr<uint8_t>(valid), as_ptr<int64_t>(err),
                   as_stream(stream));
    check_hip("gbk_decode");
  });
}
