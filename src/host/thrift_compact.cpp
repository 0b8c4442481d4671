// Generic Thrift compact-protocol reader (host).
//
// Reference parity: NativeParquetJni.cpp:33-35,692-719 parses Parquet footers
// with TCompactProtocol directly. Here the wire decoding is a generic C++
// walker that materializes structs as {field_id: value} Python dicts; the
// Parquet schema semantics live in spark_rapids_jni_amd/parquet.py.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

// Spark murmur3 for int64 (host mirror of src/gpu/srj_common.hpp mm3_*:
// two 4-byte little-endian blocks per long) — the BASELINE config[0]
// "plumbing" path and a host/GPU cross-check.
inline uint32_t h_rotl32(uint32_t x, int r) { return (x << r) | (x >> (32 - r)); }
inline uint32_t h_mm3_mix_k1(uint32_t k1) {
  k1 *= 0xcc9e2d51u;
  k1 = h_rotl32(k1, 15);
  k1 *= 0x1b873593u;
  return k1;
}
inline uint32_t h_mm3_mix_h1(uint32_t h1, uint32_t k1) {
  h1 ^= k1;
  h1 = h_rotl32(h1, 13);
  return h1 * 5u + 0xe6546b64u;
}
inline uint32_t h_mm3_fmix(uint32_t h1, uint32_t len) {
  h1 ^= len;
  h1 ^= h1 >> 16;
  h1 *= 0x85ebca6bu;
  h1 ^= h1 >> 13;
  h1 *= 0xc2b2ae35u;
  h1 ^= h1 >> 16;
  return h1;
}

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
        for (uint64_t i = 0; i < n; ++i) {
          // list bool elements are full bytes (1=true), unlike struct fields
          if (etype == 1 || etype == 2)
            out.append(py::bool_(byte() == 1));
          else
            out.append(value(etype));
        }
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

// --------------------------------------------------------------------------
// Typed parse + writer (footer re-serialization, reference
// NativeParquetJni.cpp:692-719 rewrites pruned footers with TCompactProtocol).
// Typed tree: struct -> {id: (wire_type, value)}; list/set -> (etype, [v..]);
// map -> (ktype, vtype, [(k, v)..]); bool normalized to wire_type 1.
// --------------------------------------------------------------------------

struct TypedReader : Reader {
  py::object tvalue(int type) {
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
      case 10: {
        uint8_t h = byte();
        int etype = h & 0x0F;
        uint64_t n = h >> 4;
        if (n == 15) n = varint();
        py::list out;
        for (uint64_t i = 0; i < n; ++i) {
          if (etype == 1 || etype == 2)
            out.append(py::bool_(byte() == 1));
          else
            out.append(tvalue(etype));
        }
        return py::make_tuple(etype, out);
      }
      case 11: {
        uint64_t n = varint();
        int ktype = 0, vtype = 0;
        py::list pairs;
        if (n > 0) {
          uint8_t kv = byte();
          ktype = kv >> 4;
          vtype = kv & 0x0F;
          for (uint64_t i = 0; i < n; ++i) {
            py::object k = tvalue(ktype);
            pairs.append(py::make_tuple(k, tvalue(vtype)));
          }
        }
        return py::make_tuple(ktype, vtype, pairs);
      }
      case 12: return tstrct();
      default:
        throw std::runtime_error("thrift: unknown type " + std::to_string(type));
    }
  }

  py::dict tstrct() {
    py::dict out;
    int16_t last_id = 0;
    while (true) {
      uint8_t h = byte();
      if (h == 0) return out;
      int type = h & 0x0F;
      int delta = h >> 4;
      int16_t id = delta ? (int16_t)(last_id + delta) : (int16_t)zigzag();
      last_id = id;
      int norm = type == 2 ? 1 : type;  // bool value lives in the nibble
      out[py::int_(id)] = py::make_tuple(norm, tvalue(type));
    }
  }
};

struct Writer {
  std::string out;

  void byte(uint8_t b) { out.push_back((char)b); }

  void varint(uint64_t v) {
    while (v >= 0x80) {
      byte((uint8_t)(v | 0x80));
      v >>= 7;
    }
    byte((uint8_t)v);
  }

  void zigzag(int64_t v) { varint(((uint64_t)v << 1) ^ (uint64_t)(v >> 63)); }

  void value(int type, py::handle v) {
    switch (type) {
      case 1:
      case 2: return;  // bool encoded in the type nibble / element header
      case 3: byte((uint8_t)(int8_t)py::cast<int64_t>(v)); return;
      case 4:
      case 5:
      case 6: zigzag(py::cast<int64_t>(v)); return;
      case 7: {
        double d = py::cast<double>(v);
        char buf[8];
        memcpy(buf, &d, 8);
        out.append(buf, 8);
        return;
      }
      case 8: {
        std::string s = py::cast<std::string>(v);
        varint(s.size());
        out += s;
        return;
      }
      case 9:
      case 10: {
        auto t = py::cast<py::tuple>(v);
        int etype = py::cast<int>(t[0]);
        auto elems = py::cast<py::list>(t[1]);
        size_t n = elems.size();
        if (n < 15) {
          byte((uint8_t)((n << 4) | etype));
        } else {
          byte((uint8_t)(0xF0 | etype));
          varint(n);
        }
        for (auto e : elems) {
          // bool list elements carry the value as a 1/2 byte
          if (etype == 1 || etype == 2)
            byte(py::cast<bool>(e) ? 1 : 2);
          else
            value(etype, e);
        }
        return;
      }
      case 11: {
        auto t = py::cast<py::tuple>(v);
        int ktype = py::cast<int>(t[0]), vtype = py::cast<int>(t[1]);
        auto pairs = py::cast<py::list>(t[2]);
        varint(pairs.size());
        if (pairs.size() > 0) {
          byte((uint8_t)((ktype << 4) | vtype));
          for (auto pr : pairs) {
            auto kv = py::cast<py::tuple>(pr);
            value(ktype, kv[0]);
            value(vtype, kv[1]);
          }
        }
        return;
      }
      case 12: strct(py::cast<py::dict>(v)); return;
      default:
        throw std::runtime_error("thrift write: unknown type " +
                                 std::to_string(type));
    }
  }

  void strct(py::dict d) {
    int16_t last_id = 0;
    for (auto item : d) {
      int16_t id = (int16_t)py::cast<int64_t>(item.first);
      auto tv = py::cast<py::tuple>(item.second);
      int type = py::cast<int>(tv[0]);
      py::handle v = tv[1];
      int wire = type;
      if (type == 1 || type == 2) wire = py::cast<bool>(v) ? 1 : 2;
      int delta = id - last_id;
      if (delta >= 1 && delta <= 15) {
        byte((uint8_t)((delta << 4) | wire));
      } else {
        byte((uint8_t)wire);
        zigzag(id);
      }
      last_id = id;
      value(wire, v);
    }
    byte(0);  // STOP
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
  m.def("thrift_parse_typed",
        [](py::bytes data, size_t offset) {
          char* buf;
          py::ssize_t n;
          if (PyBytes_AsStringAndSize(data.ptr(), &buf, &n) != 0)
            throw std::runtime_error("bad bytes");
          TypedReader r;
          r.p = reinterpret_cast<const uint8_t*>(buf);
          r.len = (size_t)n;
          r.pos = offset;
          py::dict d = r.tstrct();
          return py::make_tuple(d, r.pos);
        },
        py::arg("data"), py::arg("offset") = 0,
        "Parse one struct keeping wire types: {id: (type, value)}");
  m.def("thrift_write",
        [](py::dict d) {
          Writer w;
          w.strct(d);
          return py::bytes(w.out);
        },
        "Serialize a typed struct tree back to compact protocol");
  // multithreaded scatter-memcpy for the parquet page staging path: copies
  // many (src, dst_off, len) blobs into one pinned buffer in parallel — a
  // single-threaded Python loop runs at ~20 GB/s, this saturates host DRAM.
  m.def("copy_blobs",
        [](uintptr_t dst,
           std::vector<std::tuple<uintptr_t, uint64_t, uint64_t>> parts,
           int nthreads) {
          py::gil_scoped_release rel;
          if (nthreads < 1) nthreads = 1;
          uint64_t total = 0;
          for (auto& p : parts) total += std::get<2>(p);
          uint64_t per = (total + nthreads - 1) / nthreads;
          // assign whole blobs to threads by running-byte ranges; large
          // blobs are split at stripe boundaries
          std::vector<std::thread> ts;
          for (int t = 0; t < nthreads; ++t) {
            uint64_t lo = per * t, hi = per * (t + 1);
            ts.emplace_back([&, lo, hi]() {
              uint64_t pos = 0;
              for (auto& p : parts) {
                uint64_t len = std::get<2>(p);
                uint64_t a = pos, b = pos + len;
                pos = b;
                uint64_t s0 = a < lo ? lo : a;
                uint64_t s1 = b > hi ? hi : b;
                if (s0 >= s1) continue;
                std::memcpy(reinterpret_cast<char*>(dst) + std::get<1>(p) +
                                (s0 - a),
                            reinterpret_cast<const char*>(std::get<0>(p)) +
                                (s0 - a),
                            s1 - s0);
              }
            });
          }
          for (auto& t : ts) t.join();
        },
        py::arg("dst"), py::arg("parts"), py::arg("nthreads") = 8);
  m.def("murmur3_long_host",
        [](uintptr_t data, int64_t n, uint32_t seed, uintptr_t out) {
          const int64_t* v = reinterpret_cast<const int64_t*>(data);
          int32_t* o = reinterpret_cast<int32_t*>(out);
          for (int64_t i = 0; i < n; ++i) {
            uint32_t lo = (uint32_t)v[i];
            uint32_t hi = (uint32_t)(((uint64_t)v[i]) >> 32);
            uint32_t h1 = h_mm3_mix_h1(seed, h_mm3_mix_k1(lo));
            h1 = h_mm3_mix_h1(h1, h_mm3_mix_k1(hi));
            o[i] = (int32_t)h_mm3_fmix(h1, 8);
          }
        },
        "Spark murmur3 of an int64 array on the host (config[0] plumbing)");
}
