// Native (non-Python) client for the dynamo_amd request plane.
//
// Role parity with the reference's C bindings (ai-dynamo/dynamo
// lib/bindings/c): embed a worker client in any process without a Python
// runtime. Speaks the two-part codec (runtime/codec.py / the reference's
// TwoPartCodec, two_part.rs:10-60): 24-byte little-endian prefix
// {header_len, body_len, checksum} + JSON header + msgpack body. Sends
// the UNCHECKED checksum sentinel (all-ones; TCP already guarantees
// integrity) so no xxh3 dependency is needed.
//
// Header-only, blocking sockets, no external dependencies.
#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdint>
#include <cstring>
#include <functional>
#include <stdexcept>
#include <string>
#include <vector>

namespace dynamo_client {

constexpr uint64_t kUnchecked = 0xFFFFFFFFFFFFFFFFull;

// ---- minimal msgpack (the subset the worker contract uses) --------------
inline void mp_uint(std::string& o, uint64_t v) {
  if (v < 128) {
    o.push_back((char)v);
  } else if (v <= 0xFFFF) {
    o.push_back((char)0xCD);
    uint16_t be = htons((uint16_t)v);
    o.append((char*)&be, 2);
  } else if (v <= 0xFFFFFFFFull) {
    o.push_back((char)0xCE);
    uint32_t be = htonl((uint32_t)v);
    o.append((char*)&be, 4);
  } else {
    o.push_back((char)0xCF);
    for (int i = 7; i >= 0; i--) o.push_back((char)(v >> (8 * i)));
  }
}

inline void mp_str(std::string& o, const std::string& s) {
  if (s.size() < 32) {
    o.push_back((char)(0xA0 | s.size()));
  } else {
    o.push_back((char)0xD9);
    o.push_back((char)s.size());
  }
  o.append(s);
}

inline void mp_map_head(std::string& o, unsigned n) {
  o.push_back((char)(0x80 | n));   // fixmap (n < 16)
}

inline void mp_array_head(std::string& o, size_t n) {
  if (n < 16) {
    o.push_back((char)(0x90 | n));
  } else {
    o.push_back((char)0xDC);
    uint16_t be = htons((uint16_t)n);
    o.append((char*)&be, 2);
  }
}

inline void mp_bool(std::string& o, bool v) {
  o.push_back((char)(v ? 0xC3 : 0xC2));
}

// decode helpers: skip any value; extract int arrays and strings by key
struct MpReader {
  const uint8_t* p;
  const uint8_t* end;

  uint64_t be(int n) {
    uint64_t v = 0;
    for (int i = 0; i < n; i++) v = (v << 8) | *p++;
    return v;
  }

  int64_t read_int() {
    uint8_t t = *p++;
    if (t < 0x80) return t;
    if (t >= 0xE0) return (int8_t)t;
    switch (t) {
      case 0xCC: return (int64_t)be(1);
      case 0xCD: return (int64_t)be(2);
      case 0xCE: return (int64_t)be(4);
      case 0xCF: return (int64_t)be(8);
      case 0xD0: return (int8_t)be(1);
      case 0xD1: return (int16_t)be(2);
      case 0xD2: return (int32_t)be(4);
      case 0xD3: return (int64_t)be(8);
      default: throw std::runtime_error("mp: not an int");
    }
  }

  std::string read_str() {
    uint8_t t = *p++;
    size_t n;
    if ((t & 0xE0) == 0xA0) n = t & 0x1F;
    else if (t == 0xD9) n = be(1);
    else if (t == 0xDA) n = be(2);
    else throw std::runtime_error("mp: not a str");
    std::string s((const char*)p, n);
    p += n;
    return s;
  }

  size_t read_map_head() {
    uint8_t t = *p++;
    if ((t & 0xF0) == 0x80) return t & 0x0F;
    if (t == 0xDE) return be(2);
    throw std::runtime_error("mp: not a map");
  }

  size_t read_array_head() {
    uint8_t t = *p++;
    if ((t & 0xF0) == 0x90) return t & 0x0F;
    if (t == 0xDC) return be(2);
    if (t == 0xDD) return be(4);
    throw std::runtime_error("mp: not an array");
  }

  void skip() {
    uint8_t t = *p;
    if (t < 0x80 || t >= 0xE0) { p++; return; }              // fixint
    if ((t & 0xE0) == 0xA0) { size_t n = t & 0x1F; p += 1 + n; return; }
    if ((t & 0xF0) == 0x90) {                                 // fixarray
      size_t n = t & 0x0F; p++;
      for (size_t i = 0; i < n; i++) skip();
      return;
    }
    if ((t & 0xF0) == 0x80) {                                 // fixmap
      size_t n = t & 0x0F; p++;
      for (size_t i = 0; i < n; i++) { skip(); skip(); }
      return;
    }
    switch (t) {
      case 0xC0: case 0xC2: case 0xC3: p++; return;           // nil/bool
      case 0xCC: case 0xD0: p += 2; return;
      case 0xCD: case 0xD1: p += 3; return;
      case 0xCE: case 0xD2: case 0xCA: p += 5; return;
      case 0xCF: case 0xD3: case 0xCB: p += 9; return;
      case 0xD9: { p++; size_t n = be(1); p += n; return; }
      case 0xDA: { p++; size_t n = be(2); p += n; return; }
      case 0xC4: { p++; size_t n = be(1); p += n; return; }
      case 0xC5: { p++; size_t n = be(2); p += n; return; }
      case 0xDC: { p++; size_t n = be(2);
                   for (size_t i = 0; i < n; i++) skip(); return; }
      case 0xDE: { p++; size_t n = be(2);
                   for (size_t i = 0; i < n; i++) { skip(); skip(); }
                   return; }
      default: throw std::runtime_error("mp: unsupported type tag");
    }
  }
};

// ---- frames --------------------------------------------------------------
inline std::string encode_frame(const std::string& header_json,
                                const std::string& body) {
  std::string out;
  uint64_t hl = header_json.size(), bl = body.size(), cs = kUnchecked;
  out.append((char*)&hl, 8);   // little-endian hosts (x86_64)
  out.append((char*)&bl, 8);
  out.append((char*)&cs, 8);
  out += header_json;
  out += body;
  return out;
}

class Client {
 public:
  // address "host:port"
  explicit Client(const std::string& address) {
    auto colon = address.rfind(':');
    std::string host = address.substr(0, colon);
    std::string port = address.substr(colon + 1);
    struct addrinfo hints {};
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    if (getaddrinfo(host.c_str(), port.c_str(), &hints, &res) != 0)
      throw std::runtime_error("resolve failed: " + address);
    fd_ = socket(res->ai_family, res->ai_socktype, 0);
    if (fd_ < 0 || connect(fd_, res->ai_addr, res->ai_addrlen) != 0) {
      freeaddrinfo(res);
      throw std::runtime_error("connect failed: " + address);
    }
    freeaddrinfo(res);
  }

  ~Client() {
    if (fd_ >= 0) close(fd_);
  }

  // Streamed generate: calls on_tokens for each chunk's token_ids; returns
  // the total token count. finish/error handling via exceptions.
  size_t generate(const std::string& endpoint,
                  const std::vector<int64_t>& token_ids, int max_tokens,
                  const std::function<void(const std::vector<int64_t>&)>&
                      on_tokens) {
    int rid = rid_++;
    std::string header = "{\"type\":\"req\",\"rid\":" + std::to_string(rid) +
                         ",\"endpoint\":\"" + endpoint + "\"}";
    std::string body;
    mp_map_head(body, 3);
    mp_str(body, "request_id");
    mp_str(body, "cpp-" + std::to_string(rid));
    mp_str(body, "token_ids");
    mp_array_head(body, token_ids.size());
    for (auto t : token_ids) mp_uint(body, (uint64_t)t);
    mp_str(body, "stop_conditions");
    mp_map_head(body, 2);
    mp_str(body, "max_tokens");
    mp_uint(body, (uint64_t)max_tokens);
    mp_str(body, "ignore_eos");
    mp_bool(body, true);
    send_all(encode_frame(header, body));

    size_t total = 0;
    while (true) {
      auto [h, b] = read_frame();
      if (h.find("\"error\"") != std::string::npos &&
          h.find("null") == std::string::npos)
        throw std::runtime_error("endpoint error: " + h);
      bool final = h.find("\"final\":true") != std::string::npos ||
                   h.find("\"final\": true") != std::string::npos;
      if (!b.empty() && b[0] != (char)0xC0) {   // non-nil body
        MpReader r{(const uint8_t*)b.data(),
                   (const uint8_t*)b.data() + b.size()};
        size_t n = r.read_map_head();
        for (size_t i = 0; i < n; i++) {
          std::string key = r.read_str();
          if (key == "token_ids") {
            size_t m = r.read_array_head();
            std::vector<int64_t> toks(m);
            for (size_t j = 0; j < m; j++) toks[j] = r.read_int();
            total += m;
            on_tokens(toks);
          } else {
            r.skip();
          }
        }
      }
      if (final) break;
    }
    return total;
  }

 private:
  void send_all(const std::string& data) {
    size_t off = 0;
    while (off < data.size()) {
      ssize_t n = ::send(fd_, data.data() + off, data.size() - off, 0);
      if (n <= 0) throw std::runtime_error("send failed");
      off += (size_t)n;
    }
  }

  void recv_all(void* buf, size_t n) {
    size_t off = 0;
    while (off < n) {
      ssize_t r = ::recv(fd_, (char*)buf + off, n - off, 0);
      if (r <= 0) throw std::runtime_error("connection closed");
      off += (size_t)r;
    }
  }

  std::pair<std::string, std::string> read_frame() {
    uint64_t prefix[3];
    recv_all(prefix, 24);
    std::string h(prefix[0], '\0'), b(prefix[1], '\0');
    recv_all(h.data(), h.size());
    recv_all(b.data(), b.size());
    return {h, b};
  }

  int fd_ = -1;
  int rid_ = 1;
};

}  // namespace dynamo_client
