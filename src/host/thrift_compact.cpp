// Generic Thrift compact-protocol reader (host).
//
// Reference parity: NativeParquetJni.cpp:33-35,692-719 parses Parquet footers
// with TCompactProtocol directly. Here the wire decoding is a generic C++
// walker that materializes structs as {field_id: value} Python dicts; the
// Parquet schema semantics live in spark_rapids_jni_amd/parquet.py.
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>
#include <string>

namespace py = pybind11;

namespace {

struct Reader {
  const uint8_t* p;
  size_t len;
  size_t pos = 0;

  uint8_t byte() {
    if (pos >= len) throw std::runtime_error("thrift: truncated");
    return p[pos++];
  }

  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (true) {
      uint8_t b = byte();
      v |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift > 63) throw std::runtime_error("thrift: varint too long");
    }
  }

  int64_t zigzag() {
    uint64_t v = varint();
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
  }

  py::bytes binary() {
    uint64_t n = varint();
    if (pos + n > len) throw std::runtime_error("thrift: bad binary len");
    py::bytes b(reinterpret_cast<const char*>(p + pos), n);
    pos += n;
    return b;
  }

  py::object value(int type) {
    switch (type) {
      case 1: return py::bool_(true);
      case 2: return py::bool_(false);
      case 3: return py::int_((int64_t)(int8_t)byte());
      case 4:
      case 5:
      case 6: return py::int_(zigzag());
      case 7: {
        if (pos + 8 > len) throw std::runtime_error("thrift: bad double");
        double d;
        memcpy(&d, p + pos, 8);
        pos += 8;
        return py::float_(d);
      }
      case 8: return binary();
      case 9:
      case 10: {  // list / set
        uint8_t h = byte();
        int etype = h & 0x0F;
        uint64_t n = h >> 4;
        if (n == 15) n = varint();
        py::list out;
        for (uint64_t i = 0; i < n; ++i) out.append(value(etype));
        return out;
      }
      case 11: {  // map
        uint64_t n = varint();
        py::dict out;
        if (n > 0) {
          uint8_t kv = byte();
          int ktype = kv >> 4, vtype = kv & 0x0F;
          for (uint64_t i = 0; i < n; ++i) {
            py::object k = value(ktype);
            out[k] = value(vtype);
          }
        }
        return out;
      }
      case 12: return strct();
      default:
        throw std::runtime_error("thrift: unknown type " + std::to_string(type));
    }
  }

  py::dict strct() {
    py::dict out;
    int16_t last_id = 0;
    while (true) {
      uint8_t h = byte();
      if (h == 0) return out;  // STOP
      int type = h & 0x0F;
      int delta = h >> 4;
      int16_t id = delta ? (int16_t)(last_id + delta) : (int16_t)zigzag();
      last_id = id;
      out[py::int_(id)] = value(type);
    }
  }
};

}  // namespace

void register_thrift(py::module_& m) {
  m.def("thrift_parse",
        [](py::bytes data, size_t offset) {
          char* buf;
          py::ssize_t n;
          if (PyBytes_AsStringAndSize(data.ptr(), &buf, &n) != 0)
            throw std::runtime_error("bad bytes");
          Reader r{reinterpret_cast<const uint8_t*>(buf), (size_t)n, offset};
          py::dict d = r.strct();
          return py::make_tuple(d, r.pos);
        },
        py::arg("data"), py::arg("offset") = 0,
        "Parse one compact-protocol struct; returns ({field_id: value}, end)");
}
