// Bindings for JSON ops (Java API parity: JSONUtils.java).
#include "srj_bind.hpp"

extern "C" {
void srj_get_json_object(const void*, int64_t, const void*, const char*, int32_t,
                         int32_t, int32_t*, const int32_t*, char*, uint8_t*,
                         hipStream_t);
void srj_get_json_multi(const void*, int64_t, const void*, const char*,
                        const int32_t*, const int32_t*, int32_t, int32_t,
                        const void*, hipStream_t);
void srj_json_map_count(const void*, int64_t, int32_t*, hipStream_t);
void srj_json_map_entry_lens(const void*, int64_t, const int32_t*, int32_t*,
                             int32_t*, hipStream_t);
void srj_json_map_write(const void*, int64_t, const int32_t*, const int32_t*,
                        const int32_t*, char*, char*, uint8_t*, hipStream_t);
}

void register_json(py::module_& m) {
  m.def("get_json_object", [](uintptr_t in, int64_t n, uintptr_t instrs,
                              uintptr_t keychars, int32_t ninstr, int32_t phase,
                              uintptr_t lens, uintptr_t offsets, uintptr_t chars,
                              uintptr_t valid, uintptr_t stream) {
    srj_get_json_object(as_ptr<void>(in), n, as_ptr<void>(instrs),
                        as_ptr<char>(keychars), ninstr, phase,
                        as_ptr<int32_t>(lens), as_ptr<int32_t>(offsets),
                        as_ptr<char>(chars), as_ptr<uint8_t>(valid),
                        as_stream(stream));
    check_hip("get_json_object");
  });
  m.def("get_json_multi", [](uintptr_t in, int64_t n, uintptr_t instrs,
                             uintptr_t keychars, uintptr_t path_off,
                             uintptr_t path_len, int32_t npaths, int32_t phase,
                             uintptr_t outs, uintptr_t stream) {
    srj_get_json_multi(as_ptr<void>(in), n, as_ptr<void>(instrs),
                       as_ptr<char>(keychars), as_ptr<int32_t>(path_off),
                       as_ptr<int32_t>(path_len), npaths, phase,
                       as_ptr<void>(outs), as_stream(stream));
    check_hip("get_json_multi");
  });
  m.def("json_map_count", [](uintptr_t in, int64_t n, uintptr_t counts,
                             uintptr_t stream) {
    srj_json_map_count(as_ptr<void>(in), n, as_ptr<int32_t>(counts),
                       as_stream(stream));
    check_hip("json_map_count");
  });
  m.def("json_map_entry_lens", [](uintptr_t in, int64_t n, uintptr_t eoffs,
                                  uintptr_t klens, uintptr_t vlens,
                                  uintptr_t stream) {
    srj_json_map_entry_lens(as_ptr<void>(in), n, as_ptr<int32_t>(eoffs),
                            as_ptr<int32_t>(klens), as_ptr<int32_t>(vlens),
                            as_stream(stream));
    check_hip("json_map_entry_lens");
  });
  m.def("json_map_write", [](uintptr_t in, int64_t n, uintptr_t eoffs,
                             uintptr_t koffs, uintptr_t voffs, uintptr_t kchars,
                             uintptr_t vchars, uintptr_t valid, uintptr_t stream) {
    srj_json_map_write(as_ptr<void>(in), n, as_ptr<int32_t>(eoffs),
                       as_ptr<int32_t>(koffs), as_ptr<int32_t>(voffs),
                       as_ptr<char>(kchars), as_ptr<char>(vchars),
                       as_ptr<uint8_t>(valid), as_stream(stream));
    check_hip("json_map_write");
  });
}
