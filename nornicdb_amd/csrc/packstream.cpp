// Native PackStream codec (Bolt wire format), CPU-side C++.
//
// The reference's hot serialization path is compiled Go
// (pkg/bolt/packstream.go); this is the rebuild's native equivalent,
// exposed through the same _C extension and used by
// nornicdb_amd/bolt/packstream.py with a pure-python fallback.

#include <torch/extension.h>

#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

struct Packer {
  std::string out;
  py::handle structure_cls;

  void put_u8(uint8_t b) { out.push_back((char)b); }
  void put_be(const void* p, size_t n) {
    const uint8_t* b = (const uint8_t*)p;
    for (size_t i = 0; i < n; ++i) out.push_back((char)b[n - 1 - i]);
  }

  void pack_int(long long v) {
    if (v >= -16 && v < 128) {
      int8_t b = (int8_t)v;
      out.push_back((char)b);
    } else if (v >= -128 && v < 128) {
      put_u8(0xC8);
      int8_t b = (int8_t)v;
      out.push_back((char)b);
    } else if (v >= -32768 && v < 32768) {
      put_u8(0xC9);
      int16_t b = (int16_t)v;
      put_be(&b, 2);
    } else if (v >= -2147483648LL && v < 2147483648LL) {
      put_u8(0xCA);
      int32_t b = (int32_t)v;
      put_be(&b, 4);
    } else {
      put_u8(0xCB);
      put_be(&v, 8);
    }
  }

  void pack_len(size_t n, uint8_t tiny, uint8_t m8, uint8_t m16, uint8_t m32) {
    if (tiny != 0xFF && n < 0x10) {
      put_u8(tiny + (uint8_t)n);
    } else if (n < 0x100) {
      put_u8(m8);
      put_u8((uint8_t)n);
    } else if (n < 0x10000) {
      put_u8(m16);
      uint16_t b = (uint16_t)n;
      put_be(&b, 2);
    } else {
      put_u8(m32);
      uint32_t b = (uint32_t)n;
      put_be(&b, 4);
    }
  }

  void pack(py::handle v) {
    if (v.is_none()) {
      put_u8(0xC0);
      return;
    }
    if (py::isinstance<py::bool_>(v)) {
      put_u8(v.cast<bool>() ? 0xC3 : 0xC2);
      return;
    }
    if (py::isinstance<py::int_>(v)) {
      pack_int(v.cast<long long>());
      return;
    }
    if (py::isinstance<py::float_>(v)) {
      put_u8(0xC1);
      double d = v.cast<double>();
      put_be(&d, 8);
      return;
    }
    if (py::isinstance<py::str>(v)) {
      std::string s = v.cast<std::string>();
      pack_len(s.size(), 0x80, 0xD0, 0xD1, 0xD2);
      out += s;
      return;
    }
    if (py::isinstance<py::bytes>(v) || py::isinstance<py::bytearray>(v)) {
      std::string s = v.cast<std::string>();
      pack_len(s.size(), 0xFF, 0xCC, 0xCD, 0xCE);
      out += s;
      return;
    }
    if (py::isinstance<py::list>(v) || py::isinstance<py::tuple>(v)) {
      py::sequence seq = v.cast<py::sequence>();
      pack_len(seq.size(), 0x90, 0xD4, 0xD5, 0xD6);
      for (auto item : seq) pack(item);
      return;
    }
    if (py::isinstance<py::dict>(v)) {
      py::dict d = v.cast<py::dict>();
      pack_len(d.size(), 0xA0, 0xD8, 0xD9, 0xDA);
      for (auto kv : d) {
        pack(py::str(kv.first));
        pack(kv.second);
      }
      return;
    }
    // Structure duck-typing: .tag int, .fields list
    if (py::hasattr(v, "tag") && py::hasattr(v, "fields")) {
      py::sequence fields = v.attr("fields").cast<py::sequence>();
      size_t n = fields.size();
      if (n >= 0x10) throw std::runtime_error("struct too large");
      put_u8(0xB0 + (uint8_t)n);
      put_u8((uint8_t)v.attr("tag").cast<long long>());
      for (auto f : fields) pack(f);
      return;
    }
    throw std::runtime_error(
        std::string("cannot pack ") +
        py::str(v.get_type()).cast<std::string>());
  }
};

struct Unpacker {
  const uint8_t* p;
  size_t n;
  size_t i = 0;
  py::object structure_factory;

  uint8_t u8() {
    if (i >= n) throw std::runtime_error("truncated");
    return p[i++];
  }
  uint64_t be(size_t k) {
    if (i + k > n) throw std::runtime_error("truncated");
    uint64_t v = 0;
    for (size_t j = 0; j < k; ++j) v = (v << 8) | p[i++];
    return v;
  }
  py::object take_str(size_t k) {
    if (i + k > n) throw std::runtime_error("truncated");
    py::object s = py::str(std::string((const char*)p + i, k));
    i += k;
    return s;
  }
  py::object take_bytes(size_t k) {
    if (i + k > n) throw std::runtime_error("truncated");
    py::object b = py::bytes(std::string((const char*)p + i, k));
    i += k;
    return b;
  }
  py::object list_of(size_t k) {
    py::list l(k);
    for (size_t j = 0; j < k; ++j) l[j] = unpack();
    return l;
  }
  py::object map_of(size_t k) {
    py::dict d;
    for (size_t j = 0; j < k; ++j) {
      py::object key = unpack();
      d[key] = unpack();
    }
    return d;
  }
  py::object struct_of(size_t k) {
    uint8_t tag = u8();
    py::list fields(k);
    for (size_t j = 0; j < k; ++j) fields[j] = unpack();
    return structure_factory((int)tag, fields);
  }

  py::object unpack() {
    uint8_t m = u8();
    if (m <= 0x7F) return py::int_((int)m);
    if (m >= 0xF0) return py::int_((int)m - 256);
    if (m >= 0x80 && m <= 0x8F) return take_str(m & 0x0F);
    if (m >= 0x90 && m <= 0x9F) return list_of(m & 0x0F);
    if (m >= 0xA0 && m <= 0xAF) return map_of(m & 0x0F);
    if (m >= 0xB0 && m <= 0xBF) return struct_of(m & 0x0F);
    switch (m) {
      case 0xC0: return py::none();
      case 0xC1: {
        uint64_t b = be(8);
        double d;
        std::memcpy(&d, &b, 8);
        return py::float_(d);
      }
      case 0xC2: return py::bool_(false);
      case 0xC3: return py::bool_(true);
      case 0xC8: return py::int_((int)(int8_t)be(1));
      case 0xC9: return py::int_((int)(int16_t)be(2));
      case 0xCA: return py::int_((long long)(int32_t)be(4));
      case 0xCB: return py::int_((long long)be(8));
      case 0xCC: return take_bytes(be(1));
      case 0xCD: return take_bytes(be(2));
      case 0xCE: return take_bytes(be(4));
      case 0xD0: return take_str(be(1));
      case 0xD1: return take_str(be(2));
      case 0xD2: return take_str(be(4));
      case 0xD4: return list_of(be(1));
      case 0xD5: return list_of(be(2));
      case 0xD6: return list_of(be(4));
      case 0xD8: return map_of(be(1));
      case 0xD9: return map_of(be(2));
      case 0xDA: return map_of(be(4));
      case 0xDC: return struct_of(be(1));
      case 0xDD: return struct_of(be(2));
      default:
        throw std::runtime_error("unknown marker");
    }
  }
};

}  // namespace

py::bytes ps_pack(py::object v) {
  Packer pk;
  pk.out.reserve(256);
  pk.pack(v);
  return py::bytes(pk.out);
}

py::object ps_unpack(py::buffer data, py::object structure_factory) {
  py::buffer_info info = data.request();
  Unpacker u{(const uint8_t*)info.ptr, (size_t)info.size};
  u.structure_factory = structure_factory;
  return u.unpack();
}
