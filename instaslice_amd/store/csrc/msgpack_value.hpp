// Minimal msgpack codec + dynamic value for the instaslice store daemon.
//
// The store protocol (store/netstore.py) is length-prefixed msgpack maps with
// string keys; payloads are JSON-shaped (nil/bool/int/float/str/array/map).
// This implements exactly that subset — not general msgpack (no ext types,
// no bin-keyed maps). Maps preserve insertion order, matching Python dicts,
// so objects survive Python -> C++ -> Python round trips byte-comparably.

#pragma once

#include <cstdint>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace stored {

struct Value;
using Map = std::vector<std::pair<std::string, Value>>;
using Array = std::vector<Value>;

struct Value {
  enum class T : uint8_t { Nil, Bool, Int, Float, Str, Arr, Map };
  T t = T::Nil;
  bool b = false;
  int64_t i = 0;  // all integers normalized to int64 (uint64 > INT64_MAX rejected)
  double f = 0.0;
  std::string s;
  std::shared_ptr<Array> a;  // shared_ptr keeps Value cheap to copy-on-snapshot
  std::shared_ptr<Map> m;

  Value() = default;
  static Value nil() { return Value(); }
  static Value boolean(bool v) { Value x; x.t = T::Bool; x.b = v; return x; }
  static Value integer(int64_t v) { Value x; x.t = T::Int; x.i = v; return x; }
  static Value real(double v) { Value x; x.t = T::Float; x.f = v; return x; }
  static Value str(std::string v) { Value x; x.t = T::Str; x.s = std::move(v); return x; }
  static Value arr() { Value x; x.t = T::Arr; x.a = std::make_shared<Array>(); return x; }
  static Value map() { Value x; x.t = T::Map; x.m = std::make_shared<Map>(); return x; }

  bool is_map() const { return t == T::Map; }
  bool is_arr() const { return t == T::Arr; }
  bool is_str() const { return t == T::Str; }
  bool is_nil() const { return t == T::Nil; }
  bool truthy() const {
    switch (t) {
      case T::Nil: return false;
      case T::Bool: return b;
      case T::Int: return i != 0;
      case T::Float: return f != 0.0;
      case T::Str: return !s.empty();
      case T::Arr: return a && !a->empty();
      case T::Map: return m && !m->empty();
    }
    return false;
  }

  // map access (nullptr when missing / not a map)
  const Value* find(const std::string& key) const {
    if (t != T::Map || !m) return nullptr;
    for (const auto& kv : *m)
      if (kv.first == key) return &kv.second;
    return nullptr;
  }
  Value* find(const std::string& key) {
    if (t != T::Map || !m) return nullptr;
    for (auto& kv : *m)
      if (kv.first == key) return &kv.second;
    return nullptr;
  }
  Value& setkey(const std::string& key, Value v) {
    if (t != T::Map) { t = T::Map; m = std::make_shared<Map>(); }
    for (auto& kv : *m)
      if (kv.first == key) { kv.second = std::move(v); return kv.second; }
    m->emplace_back(key, std::move(v));
    return m->back().second;
  }
  bool erase(const std::string& key) {
    if (t != T::Map || !m) return false;
    for (auto it = m->begin(); it != m->end(); ++it)
      if (it->first == key) { m->erase(it); return true; }
    return false;
  }
  const std::string& str_or(const std::string& key, const std::string& dflt) const {
    static const std::string kEmpty;
    const Value* v = find(key);
    if (v && v->t == T::Str) return v->s;
    return dflt.empty() ? kEmpty : dflt;
  }
};

// deep structural copy (maps/arrays are shared_ptr — snapshot before mutating)
inline Value deep_copy(const Value& v) {
  Value out = v;
  if (v.t == Value::T::Arr && v.a) {
    out.a = std::make_shared<Array>();
    out.a->reserve(v.a->size());
    for (const auto& e : *v.a) out.a->push_back(deep_copy(e));
  } else if (v.t == Value::T::Map && v.m) {
    out.m = std::make_shared<Map>();
    out.m->reserve(v.m->size());
    for (const auto& kv : *v.m) out.m->emplace_back(kv.first, deep_copy(kv.second));
  }
  return out;
}

inline bool deep_equal(const Value& x, const Value& y) {
  // numeric cross-type equality (int 1 == float 1.0), like Python ==
  auto numeric = [](const Value& v) { return v.t == Value::T::Int || v.t == Value::T::Float; };
  if (numeric(x) && numeric(y)) {
    double xv = x.t == Value::T::Int ? static_cast<double>(x.i) : x.f;
    double yv = y.t == Value::T::Int ? static_cast<double>(y.i) : y.f;
    return xv == yv;
  }
  if (x.t != y.t) return false;
  switch (x.t) {
    case Value::T::Nil: return true;
    case Value::T::Bool: return x.b == y.b;
    case Value::T::Int: return x.i == y.i;
    case Value::T::Float: return x.f == y.f;
    case Value::T::Str: return x.s == y.s;
    case Value::T::Arr: {
      if (!x.a || !y.a) return x.a == y.a;
      if (x.a->size() != y.a->size()) return false;
      for (size_t k = 0; k < x.a->size(); ++k)
        if (!deep_equal((*x.a)[k], (*y.a)[k])) return false;
      return true;
    }
    case Value::T::Map: {
      if (!x.m || !y.m) return x.m == y.m;
      if (x.m->size() != y.m->size()) return false;
      // Python dict equality is order-insensitive
      for (const auto& kv : *x.m) {
        const Value* o = y.find(kv.first);
        if (!o || !deep_equal(kv.second, *o)) return false;
      }
      return true;
    }
  }
  return false;
}

// ---- pack ------------------------------------------------------------------

inline void pack(const Value& v, std::string& out);

inline void pack_uint_raw(uint64_t n, std::string& out) {
  if (n < 0x80) {
    out.push_back(static_cast<char>(n));
  } else if (n <= 0xff) {
    out.push_back(static_cast<char>(0xcc));
    out.push_back(static_cast<char>(n));
  } else if (n <= 0xffff) {
    out.push_back(static_cast<char>(0xcd));
    out.push_back(static_cast<char>(n >> 8));
    out.push_back(static_cast<char>(n));
  } else if (n <= 0xffffffffULL) {
    out.push_back(static_cast<char>(0xce));
    for (int s = 24; s >= 0; s -= 8) out.push_back(static_cast<char>(n >> s));
  } else {
    out.push_back(static_cast<char>(0xcf));
    for (int s = 56; s >= 0; s -= 8) out.push_back(static_cast<char>(n >> s));
  }
}

inline void pack_int(int64_t v, std::string& out) {
  if (v >= 0) { pack_uint_raw(static_cast<uint64_t>(v), out); return; }
  if (v >= -32) {
    out.push_back(static_cast<char>(v));
  } else if (v >= INT8_MIN) {
    out.push_back(static_cast<char>(0xd0));
    out.push_back(static_cast<char>(v));
  } else if (v >= INT16_MIN) {
    out.push_back(static_cast<char>(0xd1));
    out.push_back(static_cast<char>(v >> 8));
    out.push_back(static_cast<char>(v));
  } else if (v >= INT32_MIN) {
    out.push_back(static_cast<char>(0xd2));
    for (int s = 24; s >= 0; s -= 8) out.push_back(static_cast<char>(v >> s));
  } else {
    out.push_back(static_cast<char>(0xd3));
    for (int s = 56; s >= 0; s -= 8) out.push_back(static_cast<char>(v >> s));
  }
}

inline void pack_str(const std::string& s, std::string& out) {
  size_t n = s.size();
  if (n < 32) {
    out.push_back(static_cast<char>(0xa0 | n));
  } else if (n <= 0xff) {
    out.push_back(static_cast<char>(0xd9));
    out.push_back(static_cast<char>(n));
  } else if (n <= 0xffff) {
    out.push_back(static_cast<char>(0xda));
    out.push_back(static_cast<char>(n >> 8));
    out.push_back(static_cast<char>(n));
  } else {
    out.push_back(static_cast<char>(0xdb));
    for (int s2 = 24; s2 >= 0; s2 -= 8) out.push_back(static_cast<char>(n >> s2));
  }
  out.append(s);
}

inline void pack(const Value& v, std::string& out) {
  switch (v.t) {
    case Value::T::Nil:
      out.push_back(static_cast<char>(0xc0));
      break;
    case Value::T::Bool:
      out.push_back(static_cast<char>(v.b ? 0xc3 : 0xc2));
      break;
    case Value::T::Int:
      pack_int(v.i, out);
      break;
    case Value::T::Float: {
      out.push_back(static_cast<char>(0xcb));
      uint64_t bits;
      static_assert(sizeof(bits) == sizeof(v.f), "double size");
      std::memcpy(&bits, &v.f, sizeof(bits));
      for (int s = 56; s >= 0; s -= 8) out.push_back(static_cast<char>(bits >> s));
      break;
    }
    case Value::T::Str:
      pack_str(v.s, out);
      break;
    case Value::T::Arr: {
      size_t n = v.a ? v.a->size() : 0;
      if (n < 16) {
        out.push_back(static_cast<char>(0x90 | n));
      } else if (n <= 0xffff) {
        out.push_back(static_cast<char>(0xdc));
        out.push_back(static_cast<char>(n >> 8));
        out.push_back(static_cast<char>(n));
      } else {
        out.push_back(static_cast<char>(0xdd));
        for (int s = 24; s >= 0; s -= 8) out.push_back(static_cast<char>(n >> s));
      }
      if (v.a)
        for (const auto& e : *v.a) pack(e, out);
      break;
    }
    case Value::T::Map: {
      size_t n = v.m ? v.m->size() : 0;
      if (n < 16) {
        out.push_back(static_cast<char>(0x80 | n));
      } else if (n <= 0xffff) {
        out.push_back(static_cast<char>(0xde));
        out.push_back(static_cast<char>(n >> 8));
        out.push_back(static_cast<char>(n));
      } else {
        out.push_back(static_cast<char>(0xdf));
        for (int s = 24; s >= 0; s -= 8) out.push_back(static_cast<char>(n >> s));
      }
      if (v.m)
        for (const auto& kv : *v.m) {
          pack_str(kv.first, out);
          pack(kv.second, out);
        }
      break;
    }
  }
}

// ---- unpack ----------------------------------------------------------------

struct Unpacker {
  const uint8_t* p;
  const uint8_t* end;

  explicit Unpacker(const std::string& buf)
      : p(reinterpret_cast<const uint8_t*>(buf.data())),
        end(p + buf.size()) {}

  [[noreturn]] void fail(const char* what) { throw std::runtime_error(std::string("msgpack: ") + what); }

  uint64_t take_be(int n) {
    if (end - p < n) fail("truncated int");
    uint64_t v = 0;
    for (int k = 0; k < n; ++k) v = (v << 8) | *p++;
    return v;
  }

  std::string take_bytes(size_t n) {
    if (static_cast<size_t>(end - p) < n) fail("truncated bytes");
    std::string s(reinterpret_cast<const char*>(p), n);
    p += n;
    return s;
  }

  Value next() {
    if (p >= end) fail("truncated value");
    uint8_t c = *p++;
    if (c < 0x80) return Value::integer(c);                       // pos fixint
    if (c >= 0xe0) return Value::integer(static_cast<int8_t>(c)); // neg fixint
    if ((c & 0xf0) == 0x80) return take_map(c & 0x0f);            // fixmap
    if ((c & 0xf0) == 0x90) return take_arr(c & 0x0f);            // fixarray
    if ((c & 0xe0) == 0xa0) return Value::str(take_bytes(c & 0x1f));  // fixstr
    switch (c) {
      case 0xc0: return Value::nil();
      case 0xc2: return Value::boolean(false);
      case 0xc3: return Value::boolean(true);
      case 0xc4: return Value::str(take_bytes(take_be(1)));  // bin8 -> str
      case 0xc5: return Value::str(take_bytes(take_be(2)));
      case 0xc6: return Value::str(take_bytes(take_be(4)));
      case 0xca: {  // float32
        uint32_t bits = static_cast<uint32_t>(take_be(4));
        float fv;
        std::memcpy(&fv, &bits, sizeof(fv));
        return Value::real(fv);
      }
      case 0xcb: {  // float64
        uint64_t bits = take_be(8);
        double fv;
        std::memcpy(&fv, &bits, sizeof(fv));
        return Value::real(fv);
      }
      case 0xcc: return Value::integer(static_cast<int64_t>(take_be(1)));
      case 0xcd: return Value::integer(static_cast<int64_t>(take_be(2)));
      case 0xce: return Value::integer(static_cast<int64_t>(take_be(4)));
      case 0xcf: {
        uint64_t v = take_be(8);
        if (v > static_cast<uint64_t>(INT64_MAX)) fail("uint64 overflow");
        return Value::integer(static_cast<int64_t>(v));
      }
      case 0xd0: return Value::integer(static_cast<int8_t>(take_be(1)));
      case 0xd1: return Value::integer(static_cast<int16_t>(take_be(2)));
      case 0xd2: return Value::integer(static_cast<int32_t>(take_be(4)));
      case 0xd3: return Value::integer(static_cast<int64_t>(take_be(8)));
      case 0xd9: return Value::str(take_bytes(take_be(1)));
      case 0xda: return Value::str(take_bytes(take_be(2)));
      case 0xdb: return Value::str(take_bytes(take_be(4)));
      case 0xdc: return take_arr(take_be(2));
      case 0xdd: return take_arr(take_be(4));
      case 0xde: return take_map(take_be(2));
      case 0xdf: return take_map(take_be(4));
      default: fail("unsupported type tag");
    }
  }

  Value take_arr(uint64_t n) {
    Value v = Value::arr();
    v.a->reserve(n);
    for (uint64_t k = 0; k < n; ++k) v.a->push_back(next());
    return v;
  }

  Value take_map(uint64_t n) {
    Value v = Value::map();
    v.m->reserve(n);
    for (uint64_t k = 0; k < n; ++k) {
      Value key = next();
      if (key.t != Value::T::Str) fail("non-string map key");
      Value val = next();
      v.m->emplace_back(std::move(key.s), std::move(val));
    }
    return v;
  }
};

inline Value unpack(const std::string& buf) {
  Unpacker u(buf);
  return u.next();
}

}  // namespace stored
