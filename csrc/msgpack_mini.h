// Minimal msgpack encode/decode for the GPU admission RPC
// (aigw/gpu/service.py wire protocol: 4-byte LE length + msgpack map).
// Hand-written for exactly the message shapes the fast path sends
// ({id, op:"count_batch", texts:[bin]}) and receives ({id, counts:[int]});
// a full msgpack library would be dead weight in this image.
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace aigw_fast {

class MsgpackWriter {
 public:
  std::string out;

  void map_header(uint32_t n) {
    if (n <= 15) {
      out.push_back((char)(0x80 | n));
    } else {
      out.push_back((char)0xde);
      push_be16((uint16_t)n);
    }
  }

  void array_header(uint32_t n) {
    if (n <= 15) {
      out.push_back((char)(0x90 | n));
    } else if (n <= 0xffff) {
      out.push_back((char)0xdc);
      push_be16((uint16_t)n);
    } else {
      out.push_back((char)0xdd);
      push_be32(n);
    }
  }

  void str(const char* s, size_t n) {
    if (n <= 31) {
      out.push_back((char)(0xa0 | n));
    } else if (n <= 0xff) {
      out.push_back((char)0xd9);
      out.push_back((char)n);
    } else {
      out.push_back((char)0xda);
      push_be16((uint16_t)n);
    }
    out.append(s, n);
  }
  void str(const std::string& s) { str(s.data(), s.size()); }

  void bin(const char* s, size_t n) {
    if (n <= 0xff) {
      out.push_back((char)0xc4);
      out.push_back((char)n);
    } else if (n <= 0xffff) {
      out.push_back((char)0xc5);
      push_be16((uint16_t)n);
    } else {
      out.push_back((char)0xc6);
      push_be32((uint32_t)n);
    }
    out.append(s, n);
  }

  void uint(uint64_t v) {
    if (v <= 0x7f) {
      out.push_back((char)v);
    } else if (v <= 0xff) {
      out.push_back((char)0xcc);
      out.push_back((char)v);
    } else if (v <= 0xffff) {
      out.push_back((char)0xcd);
      push_be16((uint16_t)v);
    } else if (v <= 0xffffffffULL) {
      out.push_back((char)0xce);
      push_be32((uint32_t)v);
    } else {
      out.push_back((char)0xcf);
      push_be64(v);
    }
  }

 private:
  void push_be16(uint16_t v) {
    out.push_back((char)(v >> 8));
    out.push_back((char)v);
  }
  void push_be32(uint32_t v) {
    for (int i = 3; i >= 0; --i) out.push_back((char)(v >> (8 * i)));
  }
  void push_be64(uint64_t v) {
    for (int i = 7; i >= 0; --i) out.push_back((char)(v >> (8 * i)));
  }
};

// Decoder supporting the subset the GPU service replies with: maps with
// str keys, arrays, ints (pos/neg), bin/str, nil, bool. Skips unknown
// value shapes structurally.
class MsgpackReader {
 public:
  MsgpackReader(const char* p, size_t n) : p_(p), end_(p + n) {}

  bool ok() const { return ok_; }

  // Reads a map header; returns pair count or -1.
  int map_header() {
    uint8_t b = next();
    if (!ok_) return -1;
    if ((b & 0xf0) == 0x80) return b & 0x0f;
    if (b == 0xde) return (int)be16();
    if (b == 0xdf) return (int)be32();
    return fail(), -1;
  }

  int array_header() {
    uint8_t b = next();
    if (!ok_) return -1;
    if ((b & 0xf0) == 0x90) return b & 0x0f;
    if (b == 0xdc) return (int)be16();
    if (b == 0xdd) return (int)be32();
    return fail(), -1;
  }

  bool str(std::string* out) {
    uint8_t b = next();
    if (!ok_) return false;
    size_t n;
    if ((b & 0xe0) == 0xa0) n = b & 0x1f;
    else if (b == 0xd9) n = next();
    else if (b == 0xda) n = be16();
    else if (b == 0xdb) n = be32();
    else if (b == 0xc4) n = next();
    else if (b == 0xc5) n = be16();
    else if (b == 0xc6) n = be32();
    else return fail(), false;
    if (!ok_ || (size_t)(end_ - p_) < n) return fail(), false;
    if (out) out->assign(p_, n);
    p_ += n;
    return true;
  }

  bool integer(int64_t* out) {
    uint8_t b = next();
    if (!ok_) return false;
    if (b <= 0x7f) { *out = b; return true; }
    if (b >= 0xe0) { *out = (int8_t)b; return true; }
    switch (b) {
      case 0xcc: *out = next(); return ok_;
      case 0xcd: *out = be16(); return ok_;
      case 0xce: *out = be32(); return ok_;
      case 0xcf: *out = (int64_t)be64(); return ok_;
      case 0xd0: *out = (int8_t)next(); return ok_;
      case 0xd1: *out = (int16_t)be16(); return ok_;
      case 0xd2: *out = (int32_t)be32(); return ok_;
      case 0xd3: *out = (int64_t)be64(); return ok_;
      default: return fail(), false;
    }
  }

  // Structurally skip one value of any supported type.
  bool skip() {
    uint8_t b = peek();
    if (!ok_) return false;
    if (b <= 0x7f || b >= 0xe0 || b == 0xc0 || b == 0xc2 || b == 0xc3) {
      ++p_;
      return true;
    }
    if ((b & 0xe0) == 0xa0 || b == 0xd9 || b == 0xda || b == 0xdb ||
        b == 0xc4 || b == 0xc5 || b == 0xc6)
      return str(nullptr);
    if (b >= 0xcc && b <= 0xd3) {
      int64_t v;
      return integer(&v);
    }
    if (b == 0xca) { ++p_; return adv(4); }
    if (b == 0xcb) { ++p_; return adv(8); }
    if ((b & 0xf0) == 0x90 || b == 0xdc || b == 0xdd) {
      int n = array_header();
      for (int i = 0; i < n && ok_; ++i) skip();
      return ok_;
    }
    if ((b & 0xf0) == 0x80 || b == 0xde || b == 0xdf) {
      int n = map_header();
      for (int i = 0; i < n && ok_; ++i) {
        skip();
        skip();
      }
      return ok_;
    }
    return fail(), false;
  }

  bool is_nil() {
    if (peek() == 0xc0) {
      ++p_;
      return true;
    }
    return false;
  }

 private:
  uint8_t peek() {
    if (p_ >= end_) return fail(), 0;
    return (uint8_t)*p_;
  }
  uint8_t next() {
    if (p_ >= end_) return fail(), 0;
    return (uint8_t)*p_++;
  }
  bool adv(size_t n) {
    if ((size_t)(end_ - p_) < n) return fail(), false;
    p_ += n;
    return true;
  }
  uint16_t be16() { uint16_t v = next(); return (uint16_t)((v << 8) | next()); }
  uint32_t be32() {
    uint32_t v = 0;
    for (int i = 0; i < 4; ++i) v = (v << 8) | next();
    return v;
  }
  uint64_t be64() {
    uint64_t v = 0;
    for (int i = 0; i < 8; ++i) v = (v << 8) | next();
    return v;
  }
  void fail() { ok_ = false; }

  const char* p_;
  const char* end_;
  bool ok_ = true;
};

}  // namespace aigw_fast
