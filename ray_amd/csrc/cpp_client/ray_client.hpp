// ray_amd C++ client (reference: cpp/ — the C++ worker API; here the
// MI355X-native equivalent is a msgpack-native client of the
// ClientServer, src/ray/internal/internal.h role).
//
// Single-header, no third-party deps: a minimal msgpack codec plus a
// blocking TCP RPC client speaking ray_amd's framed-msgpack protocol
// (4-byte LE length + [type, seq, method, payload]; see
// ray_amd/_core/protocol.py). Gives C++ programs:
//   - cluster KV (kv_put / kv_get / kv_del)
//   - node table / cluster resources
//   - named-actor calls with plain-data args (c_actor_msgpack)
//   - registered-task calls (c_task_msgpack)
//   - pub/sub publish
//
// Example:
//   ray::Client c("127.0.0.1", 10001);
//   c.kv_put("k", "v");
//   ray::Value r = c.actor_call("counter", "incr", {ray::Value(5)});
#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdint>
#include <cstring>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

namespace ray {

// ---------------------------------------------------------------------
// Value: a msgpack-able dynamic value
// ---------------------------------------------------------------------

struct Value {
  enum Kind { NIL, BOOL, INT, FLOAT, STR, BIN, ARR, MAP } kind = NIL;
  bool b = false;
  int64_t i = 0;
  double f = 0.0;
  std::string s;  // STR and BIN payload
  std::vector<Value> arr;
  std::vector<std::pair<Value, Value>> map;

  Value() = default;
  Value(bool v) : kind(BOOL), b(v) {}
  Value(int v) : kind(INT), i(v) {}
  Value(int64_t v) : kind(INT), i(v) {}
  Value(double v) : kind(FLOAT), f(v) {}
  Value(const char* v) : kind(STR), s(v) {}
  Value(const std::string& v) : kind(STR), s(v) {}
  static Value bin(std::string v) {
    Value x; x.kind = BIN; x.s = std::move(v); return x;
  }
  static Value array(std::vector<Value> v) {
    Value x; x.kind = ARR; x.arr = std::move(v); return x;
  }
  static Value object(std::vector<std::pair<Value, Value>> v) {
    Value x; x.kind = MAP; x.map = std::move(v); return x;
  }

  const Value* get(const std::string& key) const {
    for (auto& kv : map)
      if (kv.first.kind == STR && kv.first.s == key) return &kv.second;
    return nullptr;
  }
  bool truthy() const {
    switch (kind) {
      case BOOL: return b;
      case INT: return i != 0;
      case NIL: return false;
      default: return true;
    }
  }
  int64_t as_int() const { return kind == FLOAT ? (int64_t)f : i; }
  double as_float() const { return kind == INT ? (double)i : f; }
};

// ---------------------------------------------------------------------
// msgpack codec (the subset both sides use)
// ---------------------------------------------------------------------

namespace msgpack {

inline void put_be(std::string& o, uint64_t v, int n) {
  for (int k = n - 1; k >= 0; --k) o.push_back((char)((v >> (8 * k)) & 0xff));
}

inline void encode(const Value& v, std::string& o) {
  switch (v.kind) {
    case Value::NIL: o.push_back((char)0xc0); break;
    case Value::BOOL: o.push_back((char)(v.b ? 0xc3 : 0xc2)); break;
    case Value::INT: {
      int64_t x = v.i;
      if (x >= 0 && x < 128) o.push_back((char)x);
      else if (x < 0 && x >= -32) o.push_back((char)(int8_t)x);
      else { o.push_back((char)0xd3); put_be(o, (uint64_t)x, 8); }
      break;
    }
    case Value::FLOAT: {
      o.push_back((char)0xcb);
      uint64_t bits; std::memcpy(&bits, &v.f, 8); put_be(o, bits, 8);
      break;
    }
    case Value::STR: {
      size_t n = v.s.size();
      if (n < 32) o.push_back((char)(0xa0 | n));
      else if (n < 256) { o.push_back((char)0xd9); o.push_back((char)n); }
      else { o.push_back((char)0xda); put_be(o, n, 2); }
      o += v.s;
      break;
    }
    case Value::BIN: {
      size_t n = v.s.size();
      if (n < 256) { o.push_back((char)0xc4); o.push_back((char)n); }
      else if (n < 65536) { o.push_back((char)0xc5); put_be(o, n, 2); }
      else { o.push_back((char)0xc6); put_be(o, n, 4); }
      o += v.s;
      break;
    }
    case Value::ARR: {
      size_t n = v.arr.size();
      if (n < 16) o.push_back((char)(0x90 | n));
      else { o.push_back((char)0xdc); put_be(o, n, 2); }
      for (auto& e : v.arr) encode(e, o);
      break;
    }
    case Value::MAP: {
      size_t n = v.map.size();
      if (n < 16) o.push_back((char)(0x80 | n));
      else { o.push_back((char)0xde); put_be(o, n, 2); }
      for (auto& kv : v.map) { encode(kv.first, o); encode(kv.second, o); }
      break;
    }
  }
}

struct Reader {
  const uint8_t* p;
  const uint8_t* end;
  uint64_t be(int n) {
    if (end - p < n) throw std::runtime_error("msgpack: truncated");
    uint64_t v = 0;
    for (int k = 0; k < n; ++k) v = (v << 8) | *p++;
    return v;
  }
  std::string bytes(size_t n) {
    if ((size_t)(end - p) < n) throw std::runtime_error("msgpack: truncated");
    std::string s((const char*)p, n);
    p += n;
    return s;
  }
  Value decode() {
    if (p >= end) throw std::runtime_error("msgpack: empty");
    uint8_t t = *p++;
    Value v;
    if (t < 0x80) { v.kind = Value::INT; v.i = t; return v; }
    if (t >= 0xe0) { v.kind = Value::INT; v.i = (int8_t)t; return v; }
    if ((t & 0xf0) == 0x90) return arr(t & 0x0f);
    if ((t & 0xf0) == 0x80) return mp(t & 0x0f);
    if ((t & 0xe0) == 0xa0) { v.kind = Value::STR; v.s = bytes(t & 0x1f); return v; }
    switch (t) {
      case 0xc0: return v;
      case 0xc2: v.kind = Value::BOOL; v.b = false; return v;
      case 0xc3: v.kind = Value::BOOL; v.b = true; return v;
      case 0xc4: { size_t n = be(1); v.kind = Value::BIN; v.s = bytes(n); return v; }
      case 0xc5: { size_t n = be(2); v.kind = Value::BIN; v.s = bytes(n); return v; }
      case 0xc6: { size_t n = be(4); v.kind = Value::BIN; v.s = bytes(n); return v; }
      case 0xca: { uint32_t x = (uint32_t)be(4); float f; std::memcpy(&f, &x, 4);
                   v.kind = Value::FLOAT; v.f = f; return v; }
      case 0xcb: { uint64_t x = be(8); std::memcpy(&v.f, &x, 8);
                   v.kind = Value::FLOAT; return v; }
      case 0xcc: v.kind = Value::INT; v.i = (int64_t)be(1); return v;
      case 0xcd: v.kind = Value::INT; v.i = (int64_t)be(2); return v;
      case 0xce: v.kind = Value::INT; v.i = (int64_t)be(4); return v;
      case 0xcf: v.kind = Value::INT; v.i = (int64_t)be(8); return v;
      case 0xd0: v.kind = Value::INT; v.i = (int8_t)be(1); return v;
      case 0xd1: v.kind = Value::INT; v.i = (int16_t)be(2); return v;
      case 0xd2: v.kind = Value::INT; v.i = (int32_t)be(4); return v;
      case 0xd3: v.kind = Value::INT; v.i = (int64_t)be(8); return v;
      case 0xd9: { size_t n = be(1); v.kind = Value::STR; v.s = bytes(n); return v; }
      case 0xda: { size_t n = be(2); v.kind = Value::STR; v.s = bytes(n); return v; }
      case 0xdb: { size_t n = be(4); v.kind = Value::STR; v.s = bytes(n); return v; }
      case 0xdc: return arr(be(2));
      case 0xdd: return arr(be(4));
      case 0xde: return mp(be(2));
      case 0xdf: return mp(be(4));
      default: throw std::runtime_error("msgpack: unsupported tag");
    }
  }
  Value arr(size_t n) {
    Value v; v.kind = Value::ARR; v.arr.reserve(n);
    for (size_t k = 0; k < n; ++k) v.arr.push_back(decode());
    return v;
  }
  Value mp(size_t n) {
    Value v; v.kind = Value::MAP; v.map.reserve(n);
    for (size_t k = 0; k < n; ++k) {
      Value key = decode();
      v.map.emplace_back(std::move(key), decode());
    }
    return v;
  }
};

}  // namespace msgpack

// ---------------------------------------------------------------------
// Client
// ---------------------------------------------------------------------

class Client {
 public:
  Client(const std::string& host, int port) {
    fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    ::setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (::inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
      throw std::runtime_error("bad host " + host);
    if (::connect(fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("connect() failed");
    call("c_init", Value::object({{Value("namespace"), Value()}}));
  }
  ~Client() {
    if (fd_ >= 0) ::close(fd_);
  }

  // ---- raw RPC ----

  Value call(const std::string& method, const Value& payload) {
    int64_t seq = ++seq_;
    Value frame = Value::array(
        {Value((int64_t)0), Value(seq), Value(method), payload});
    std::string body;
    msgpack::encode(frame, body);
    std::string msg;
    uint32_t len = (uint32_t)body.size();
    char hdr[4] = {(char)(len & 0xff), (char)((len >> 8) & 0xff),
                   (char)((len >> 16) & 0xff), (char)((len >> 24) & 0xff)};
    msg.assign(hdr, 4);
    msg += body;
    send_all(msg);
    for (;;) {
      Value reply = read_frame();
      int64_t t = reply.arr.at(0).i;
      if (t == 3) continue;  // notify: ignore (no subscriptions here)
      if (reply.arr.at(1).i != seq) continue;
      if (t == 2)
        throw std::runtime_error("rpc error: " + reply.arr.at(2).s);
      return reply.arr.at(2);  // [REPLY, seq, payload]
    }
  }

  // ---- KV (GCS passthrough) ----

  void kv_put(const std::string& key, const std::string& val,
              const std::string& ns = "") {
    gcs("kv_put", Value::object({{Value("ns"), Value(ns)},
                                 {Value("key"), Value::bin(key)},
                                 {Value("value"), Value::bin(val)},
                                 {Value("overwrite"), Value(true)}}));
  }
  std::string kv_get(const std::string& key, const std::string& ns = "") {
    Value r = gcs("kv_get", Value::object({{Value("ns"), Value(ns)},
                                           {Value("key"), Value::bin(key)}}));
    return r.kind == Value::NIL ? std::string() : r.s;
  }
  void kv_del(const std::string& key, const std::string& ns = "") {
    gcs("kv_del", Value::object({{Value("ns"), Value(ns)},
                                 {Value("key"), Value::bin(key)}}));
  }

  Value node_table() { return gcs("node_table", Value::object({})); }

  // ---- named-actor + registered-task calls (plain-data args) ----

  Value actor_call(const std::string& actor_name, const std::string& method,
                   std::vector<Value> args, double timeout_s = 60.0) {
    Value r = call("c_actor_msgpack",
                   Value::object({{Value("name"), Value(actor_name)},
                                  {Value("method"), Value(method)},
                                  {Value("args"), Value::array(std::move(args))},
                                  {Value("timeout"), Value(timeout_s)}}));
    return unwrap(r);
  }

  Value task_call(const std::string& task_name, std::vector<Value> args,
                  double timeout_s = 60.0) {
    Value r = call("c_task_msgpack",
                   Value::object({{Value("name"), Value(task_name)},
                                  {Value("args"), Value::array(std::move(args))},
                                  {Value("timeout"), Value(timeout_s)}}));
    return unwrap(r);
  }

  int64_t publish(const std::string& channel, const Value& data) {
    return gcs("publish", Value::object({{Value("channel"), Value(channel)},
                                         {Value("data"), data}}))
        .as_int();
  }

 private:
  Value gcs(const std::string& method, const Value& payload) {
    return call("c_gcs", Value::object({{Value("method"), Value(method)},
                                        {Value("payload"), payload}}));
  }
  static Value unwrap(Value& r) {
    const Value* ok = r.get("ok");
    if (ok == nullptr || !ok->truthy()) {
      const Value* err = r.get("error");
      throw std::runtime_error(err ? err->s : "remote call failed");
    }
    const Value* v = r.get("value");
    return v ? *v : Value();
  }
  void send_all(const std::string& data) {
    size_t off = 0;
    while (off < data.size()) {
      ssize_t n = ::send(fd_, data.data() + off, data.size() - off, 0);
      if (n <= 0) throw std::runtime_error("send failed");
      off += (size_t)n;
    }
  }
  void recv_all(uint8_t* buf, size_t n) {
    size_t off = 0;
    while (off < n) {
      ssize_t r = ::recv(fd_, buf + off, n - off, 0);
      if (r <= 0) throw std::runtime_error("connection closed");
      off += (size_t)r;
    }
  }
  Value read_frame() {
    uint8_t hdr[4];
    recv_all(hdr, 4);
    uint32_t len = (uint32_t)hdr[0] | ((uint32_t)hdr[1] << 8) |
                   ((uint32_t)hdr[2] << 16) | ((uint32_t)hdr[3] << 24);
    std::vector<uint8_t> body(len);
    recv_all(body.data(), len);
    msgpack::Reader rd{body.data(), body.data() + len};
    return rd.decode();
  }

  int fd_ = -1;
  int64_t seq_ = 0;
};

}  // namespace ray
