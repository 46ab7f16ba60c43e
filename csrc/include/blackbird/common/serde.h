// Compact little-endian binary serialization for the RPC and coordination
// wire format. The reference rode YLT struct_pack (rpc_service.h:7-8); this
// framework uses its own ~150-line archive: explicit, dependency-free, and
// stable across compilers.
#pragma once

#include <cstdint>
#include <cstring>
#include <map>
#include <optional>
#include <string>
#include <type_traits>
#include <vector>

namespace blackbird::serde {

class Enc {
 public:
  std::string buf;

  void raw(const void* p, size_t n) { buf.append(static_cast<const char*>(p), n); }
  template <typename T>
  std::enable_if_t<std::is_arithmetic_v<T> || std::is_enum_v<T>> num(T v) {
    raw(&v, sizeof(v));
  }
  void str(const std::string& s) {
    num<uint32_t>(static_cast<uint32_t>(s.size()));
    raw(s.data(), s.size());
  }
  void bytes(const void* p, size_t n) {
    num<uint64_t>(n);
    raw(p, n);
  }
};

class Dec {
 public:
  Dec(const void* p, size_t n) : p_(static_cast<const char*>(p)), end_(p_ + n) {}
  bool ok() const { return ok_; }
  void fail() { ok_ = false; }
  size_t remaining() const { return static_cast<size_t>(end_ - p_); }

  bool raw(void* out, size_t n) {
    if (!ok_ || remaining() < n) { ok_ = false; return false; }
    std::memcpy(out, p_, n);
    p_ += n;
    return true;
  }
  template <typename T>
  std::enable_if_t<std::is_arithmetic_v<T> || std::is_enum_v<T>, T> num() {
    T v{};
    raw(&v, sizeof(v));
    return v;
  }
  std::string str() {
    uint32_t n = num<uint32_t>();
    if (!ok_ || remaining() < n) { ok_ = false; return {}; }
    std::string s(p_, n);
    p_ += n;
    return s;
  }
  std::string bytes() {
    uint64_t n = num<uint64_t>();
    if (!ok_ || remaining() < n) { ok_ = false; return {}; }
    std::string s(p_, static_cast<size_t>(n));
    p_ += n;
    return s;
  }

 private:
  const char* p_;
  const char* end_;
  bool ok_ = true;
};

// --- generic helpers: a type is serializable if it has enc(Enc&)/dec(Dec&)
template <typename T>
concept Struct = requires(const T ct, T t, Enc& e, Dec& d) {
  ct.enc(e);
  t.dec(d);
};

template <typename T>
void put(Enc& e, const T& v) {
  if constexpr (Struct<T>) v.enc(e);
  else if constexpr (std::is_same_v<T, std::string>) e.str(v);
  else e.num(v);
}

template <typename T>
void get(Dec& d, T& v) {
  if constexpr (Struct<T>) v.dec(d);
  else if constexpr (std::is_same_v<T, std::string>) v = d.str();
  else v = d.num<T>();
}

template <typename T>
void put(Enc& e, const std::vector<T>& v) {
  e.num<uint32_t>(static_cast<uint32_t>(v.size()));
  for (const auto& x : v) put(e, x);
}

template <typename T>
void get(Dec& d, std::vector<T>& v) {
  uint32_t n = d.num<uint32_t>();
  v.clear();
  // Guard against hostile/corrupt lengths: each element needs ≥1 byte. The
  // rejection must surface as a decode error, not as a valid empty vector.
  if (n > d.remaining() && n > (1u << 24)) { d.fail(); return; }
  v.reserve(n);
  for (uint32_t i = 0; i < n && d.ok(); ++i) {
    T x{};
    get(d, x);
    v.push_back(std::move(x));
  }
}

template <typename K, typename V>
void put(Enc& e, const std::map<K, V>& m) {
  e.num<uint32_t>(static_cast<uint32_t>(m.size()));
  for (const auto& [k, v] : m) { put(e, k); put(e, v); }
}

template <typename K, typename V>
void get(Dec& d, std::map<K, V>& m) {
  uint32_t n = d.num<uint32_t>();
  m.clear();
  for (uint32_t i = 0; i < n && d.ok(); ++i) {
    K k{}; V v{};
    get(d, k); get(d, v);
    m.emplace(std::move(k), std::move(v));
  }
}

template <typename T>
void put(Enc& e, const std::optional<T>& o) {
  e.num<uint8_t>(o.has_value() ? 1 : 0);
  if (o) put(e, *o);
}

template <typename T>
void get(Dec& d, std::optional<T>& o) {
  if (d.num<uint8_t>()) { T v{}; get(d, v); o = std::move(v); }
  else o.reset();
}

template <typename T>
std::string to_bytes(const T& v) {
  Enc e;
  put(e, v);
  return std::move(e.buf);
}

template <typename T>
bool from_bytes(const std::string& s, T& out) {
  Dec d(s.data(), s.size());
  get(d, out);
  return d.ok();
}

}  // namespace blackbird::serde

// BB_FIELDS(members...) declares enc/dec over the listed members (the
// variadic arguments are member expressions, valid inside the struct).
#define BB_FIELDS(...)                                       \
  void enc(::blackbird::serde::Enc& _e) const {              \
    bb_each_enc(_e, __VA_ARGS__);                            \
  }                                                          \
  void dec(::blackbird::serde::Dec& _d) {                    \
    bb_each_dec(_d, __VA_ARGS__);                            \
  }                                                          \
  template <typename... Ts>                                  \
  void bb_each_enc(::blackbird::serde::Enc& _e, const Ts&... xs) const { \
    (::blackbird::serde::put(_e, xs), ...);                  \
  }                                                          \
  template <typename... Ts>                                  \
  void bb_each_dec(::blackbird::serde::Dec& _d, Ts&... xs) { \
    (::blackbird::serde::get(_d, xs), ...);                  \
  }
