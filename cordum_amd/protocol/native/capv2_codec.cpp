// CAP v2 deterministic protobuf codec — native engine.
//
// Schema-driven: Python registers each message class's FIELDS table once
// (protocol/capv2.py `_register_native`), so the wire layout has exactly one
// source of truth and the two codecs cannot drift; tests/test_property.py
// asserts byte-identical output against the pure-Python encoder on random
// messages. Wire format notes mirror capv2.py: ascending field numbers,
// map entries sorted by key, defaults omitted, negative ints as 64-bit
// two's-complement varints (oracle: the reference's protobuf use, and
// job_hash determinism core/controlplane/scheduler/job_hash.go:15-48).
//
// Plain CPython/pybind11 module — no torch/HIP dependency; the control
// plane's host runtime loads it for the API boundary, WAL and worker wire
// paths. Built in-tree by cordum_amd/ops/__init__.build_extension.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstring>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

enum Kind : int {
  K_INT = 0,
  K_SINT64 = 1,
  K_BOOL = 2,
  K_ENUM = 3,
  K_STR = 4,
  K_BYTES = 5,
  K_DOUBLE = 6,
  K_MSG = 7,
  K_REP_STR = 8,
  K_MAP_SS = 9,
};

struct FieldSpec {
  std::string name;
  int num;
  int kind;
  std::string sub;   // registered type key for K_MSG
  py::object enum_cls;  // IntEnum class for K_ENUM (may be None)
};

struct MsgSchema {
  std::vector<FieldSpec> fields;  // sorted by num
  py::object cls;                 // Python class (factory for decode)
  std::map<int, int> by_num;      // field number -> index in fields
};

static std::map<std::string, MsgSchema>& registry() {
  static std::map<std::string, MsgSchema> r;
  return r;
}

// -- wire primitives ---------------------------------------------------------

static inline void put_varint(std::string& out, unsigned long long v) {
  while (true) {
    unsigned char b = v & 0x7F;
    v >>= 7;
    if (v) {
      out.push_back((char)(b | 0x80));
    } else {
      out.push_back((char)b);
      return;
    }
  }
}

static inline void put_tag(std::string& out, int num, int wt) {
  put_varint(out, ((unsigned long long)num << 3) | (unsigned)wt);
}

static inline unsigned long long get_varint(const unsigned char* buf, size_t n,
                                            size_t& i) {
  unsigned long long val = 0;
  int shift = 0;
  while (true) {
    if (i >= n) throw std::runtime_error("varint truncated");
    unsigned char b = buf[i++];
    val |= (unsigned long long)(b & 0x7F) << shift;
    if (!(b & 0x80)) return val;
    shift += 7;
    if (shift > 70) throw std::runtime_error("varint overflow");
  }
}

// -- encode ------------------------------------------------------------------

static void encode_msg(const std::string& key, py::handle obj, std::string& out);

static void encode_field(const FieldSpec& f, py::handle val, std::string& out) {
  switch (f.kind) {
    case K_INT:
    case K_SINT64:
    case K_ENUM: {
      long long v = PyLong_AsLongLong(PyNumber_Long(val.ptr()));
      if (v == -1 && PyErr_Occurred()) throw py::error_already_set();
      if (v == 0) return;
      put_tag(out, f.num, 0);
      put_varint(out, (unsigned long long)v);
      return;
    }
    case K_BOOL: {
      if (!py::cast<bool>(val)) return;
      put_tag(out, f.num, 0);
      put_varint(out, 1);
      return;
    }
    case K_DOUBLE: {
      double d = py::cast<double>(val);
      if (d == 0.0) return;
      put_tag(out, f.num, 1);
      char b[8];
      std::memcpy(b, &d, 8);
      out.append(b, 8);
      return;
    }
    case K_STR: {
      std::string s = py::cast<std::string>(val);
      if (s.empty()) return;
      put_tag(out, f.num, 2);
      put_varint(out, s.size());
      out += s;
      return;
    }
    case K_BYTES: {
      if (!val || val.is_none()) return;
      std::string s = py::cast<std::string>(val);
      if (s.empty()) return;
      put_tag(out, f.num, 2);
      put_varint(out, s.size());
      out += s;
      return;
    }
    case K_MSG: {
      if (val.is_none()) return;
      std::string sub;
      encode_msg(f.sub, val, sub);
      put_tag(out, f.num, 2);
      put_varint(out, sub.size());
      out += sub;
      return;
    }
    case K_REP_STR: {
      for (py::handle item : py::cast<py::list>(val)) {
        std::string s = py::cast<std::string>(item);
        put_tag(out, f.num, 2);
        put_varint(out, s.size());
        out += s;
      }
      return;
    }
    case K_MAP_SS: {
      auto d = py::cast<py::dict>(val);
      std::vector<std::pair<std::string, std::string>> items;
      items.reserve(d.size());
      for (auto kv : d)
        items.emplace_back(py::cast<std::string>(kv.first),
                           py::cast<std::string>(kv.second));
      std::sort(items.begin(), items.end());  // deterministic, like capv2.py
      for (auto& kv : items) {
        std::string entry;
        // map entries serialize BOTH fields even when empty (protobuf
        // MapEntry semantics; byte-verified in tests/test_capv2_interop.py)
        put_tag(entry, 1, 2);
        put_varint(entry, kv.first.size());
        entry += kv.first;
        put_tag(entry, 2, 2);
        put_varint(entry, kv.second.size());
        entry += kv.second;
        put_tag(out, f.num, 2);
        put_varint(out, entry.size());
        out += entry;
      }
      return;
    }
  }
  throw std::runtime_error("unknown field kind");
}

static void encode_msg(const std::string& key, py::handle obj, std::string& out) {
  auto it = registry().find(key);
  if (it == registry().end()) throw std::runtime_error("unregistered type " + key);
  for (const FieldSpec& f : it->second.fields) {
    py::object val = py::reinterpret_borrow<py::object>(obj).attr(f.name.c_str());
    encode_field(f, val, out);
  }
  // unknown fields captured by the Python decoder re-emit verbatim after
  // known fields (Go protobuf layout) so relays don't strip foreign fields
  PyObject* unk = PyObject_GetAttrString(obj.ptr(), "_unknown");
  if (unk == nullptr) {
    PyErr_Clear();
  } else {
    if (PyBytes_Check(unk) && PyBytes_GET_SIZE(unk) > 0)
      out.append(PyBytes_AS_STRING(unk), (size_t)PyBytes_GET_SIZE(unk));
    Py_DECREF(unk);
  }
}

// -- decode ------------------------------------------------------------------

static py::object decode_msg(const std::string& key, const unsigned char* buf,
                             size_t n);

static void decode_field(py::object& msg, const FieldSpec& f,
                         const unsigned char* raw, size_t rn,
                         unsigned long long varint_raw, int wt) {
  switch (f.kind) {
    case K_INT:
    case K_SINT64: {
      long long v = (long long)varint_raw;  // two's complement, like capv2.py
      msg.attr(f.name.c_str()) = py::int_(v);
      return;
    }
    case K_ENUM: {
      // mirror capv2.py _decode_into: try the IntEnum, fall back to raw int
      py::int_ raw_val((long long)varint_raw);
      if (!f.enum_cls.is_none()) {
        try {
          msg.attr(f.name.c_str()) = f.enum_cls(raw_val);
          return;
        } catch (py::error_already_set&) {
        }
      }
      msg.attr(f.name.c_str()) = raw_val;
      return;
    }
    case K_BOOL:
      msg.attr(f.name.c_str()) = py::bool_(varint_raw != 0);
      return;
    case K_DOUBLE: {
      double d;
      std::memcpy(&d, raw, 8);
      msg.attr(f.name.c_str()) = py::float_(d);
      return;
    }
    case K_STR:
      msg.attr(f.name.c_str()) = py::str(std::string((const char*)raw, rn));
      return;
    case K_BYTES:
      msg.attr(f.name.c_str()) = py::bytes((const char*)raw, rn);
      return;
    case K_MSG:
      msg.attr(f.name.c_str()) = decode_msg(f.sub, raw, rn);
      return;
    case K_REP_STR: {
      py::list lst = msg.attr(f.name.c_str());
      lst.append(py::str(std::string((const char*)raw, rn)));
      return;
    }
    case K_MAP_SS: {
      std::string k, v;
      size_t i = 0;
      while (i < rn) {
        unsigned long long tag = get_varint(raw, rn, i);
        unsigned long long ln = get_varint(raw, rn, i);
        if (i + ln > rn) throw std::runtime_error("map entry truncated");
        std::string s((const char*)raw + i, ln);
        i += ln;
        if ((tag >> 3) == 1) k = s; else v = s;
      }
      py::dict d = msg.attr(f.name.c_str());
      d[py::str(k)] = py::str(v);
      return;
    }
  }
  throw std::runtime_error("unknown field kind");
}

static py::object decode_msg(const std::string& key, const unsigned char* buf,
                             size_t n) {
  auto it = registry().find(key);
  if (it == registry().end()) throw std::runtime_error("unregistered type " + key);
  py::object msg = it->second.cls();
  size_t i = 0;
  while (i < n) {
    unsigned long long tag = get_varint(buf, n, i);
    int num = (int)(tag >> 3);
    int wt = (int)(tag & 7);
    const unsigned char* raw = nullptr;
    size_t rn = 0;
    unsigned long long vint = 0;
    if (wt == 0) {
      vint = get_varint(buf, n, i);
    } else if (wt == 2) {
      unsigned long long ln = get_varint(buf, n, i);
      if (i + ln > n) throw std::runtime_error("field truncated");
      raw = buf + i;
      rn = ln;
      i += ln;
    } else if (wt == 1) {
      if (i + 8 > n) throw std::runtime_error("i64 truncated");
      raw = buf + i;
      rn = 8;
      i += 8;
    } else if (wt == 5) {
      if (i + 4 > n) throw std::runtime_error("i32 truncated");
      raw = buf + i;
      rn = 4;
      i += 4;
    } else {
      throw std::runtime_error("bad wire type");
    }
    auto fit = it->second.by_num.find(num);
    if (fit == it->second.by_num.end())
      // unknown field (newer CAP peer): defer the whole packet to the
      // Python codec, which preserves unknown bytes for re-encode; the
      // dispatcher treats this throw as "fall back"
      throw std::runtime_error("unknown field " + std::to_string(num));
    const FieldSpec& fs = it->second.fields[fit->second];
    const int want_wt =
        (fs.kind == K_INT || fs.kind == K_SINT64 || fs.kind == K_BOOL ||
         fs.kind == K_ENUM) ? 0 : (fs.kind == K_DOUBLE ? 1 : 2);
    if (wt != want_wt)
      throw std::runtime_error("wire type mismatch for field " +
                               std::to_string(num));
    decode_field(msg, fs, raw, rn, vint, wt);
  }
  return msg;
}

// -- module ------------------------------------------------------------------

PYBIND11_MODULE(_capv2_native, m) {
  m.doc() = "CAP v2 deterministic protobuf codec (schema-driven native engine)";
  m.def("register_message",
        [](const std::string& key, py::object cls,
           std::vector<std::tuple<std::string, int, int, std::string, py::object>>
               fields) {
          MsgSchema s;
          s.cls = cls;
          for (auto& t : fields)
            s.fields.push_back({std::get<0>(t), std::get<1>(t), std::get<2>(t),
                                std::get<3>(t), std::get<4>(t)});
          std::sort(s.fields.begin(), s.fields.end(),
                    [](const FieldSpec& a, const FieldSpec& b) {
                      return a.num < b.num;
                    });
          for (size_t i = 0; i < s.fields.size(); ++i)
            s.by_num[s.fields[i].num] = (int)i;
          registry()[key] = std::move(s);
        },
        py::arg("key"), py::arg("cls"), py::arg("fields"));
  m.def("encode", [](const std::string& key, py::object obj) {
    std::string out;
    encode_msg(key, obj, out);
    return py::bytes(out);
  });
  m.def("decode", [](const std::string& key, py::bytes data) {
    std::string s = data;
    return decode_msg(key, (const unsigned char*)s.data(), s.size());
  });
}
