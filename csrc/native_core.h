// aigw native core — pybind-free hot-path scanners shared by the Python
// extension (csrc/aigw_native.cpp) and the standalone sanitizer test
// binary (csrc/native_core_test.cpp, built with -fsanitize=address,
// undefined by scripts/sanitize_native.sh — the race/memory-safety
// check the reference gets from `go test -race`).
#pragma once

#include <cstddef>
#include <cstring>
#include <string>
#include <utility>
#include <vector>

namespace aigw_core {

struct Scan {
  const char* p;
  const char* end;
  std::string model;
  int stream = 0;
  std::string text;
  bool ok = true;
  const char* base = nullptr;     // set to the body start to capture spans
  size_t model_vs = 0, model_ve = 0;  // "model" value span incl. quotes
  size_t msgs_vs = 0, msgs_ve = 0;    // top-level "messages" value span

  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  }

  // parse a JSON string; when collect != nullptr, append the unescaped value.
  // The common no-escape span is scanned with a tight byte loop (auto-
  // vectorized) and appended in bulk — per-char push_back made the scanner
  // slower than CPython's json for 16 KiB bodies.
  bool parse_string(std::string* collect) {
    if (p >= end || *p != '"') return fail();
    ++p;
    while (p < end) {
      const char* seg = p;
      while (p < end && *p != '"' && *p != '\\') ++p;
      if (collect && p != seg) collect->append(seg, p - seg);
      if (p >= end) return fail();
      unsigned char c = *p;
      if (c == '"') {
        ++p;
        return true;
      }
      if (c == '\\') {
        if (p + 1 >= end) return fail();
        char e = p[1];
        p += 2;
        if (!collect) {
          // strict validation without materializing the value
          switch (e) {
            case '"': case '\\': case '/': case 'b': case 'f':
            case 'n': case 'r': case 't':
              break;
            case 'u': {
              if (p + 4 > end) return fail();
              for (int i = 0; i < 4; ++i) {
                char h = p[i];
                if (!((h >= '0' && h <= '9') || (h >= 'a' && h <= 'f') ||
                      (h >= 'A' && h <= 'F')))
                  return fail();
              }
              p += 4;
              break;
            }
            default:
              return fail();
          }
          continue;
        }
        switch (e) {
          case '"': collect->push_back('"'); break;
          case '\\': collect->push_back('\\'); break;
          case '/': collect->push_back('/'); break;
          case 'b': collect->push_back('\b'); break;
          case 'f': collect->push_back('\f'); break;
          case 'n': collect->push_back('\n'); break;
          case 'r': collect->push_back('\r'); break;
          case 't': collect->push_back('\t'); break;
          case 'u': {
            if (p + 4 > end) return fail();
            unsigned v = 0;
            for (int i = 0; i < 4; ++i) {
              char h = p[i];
              v <<= 4;
              if (h >= '0' && h <= '9') v |= h - '0';
              else if (h >= 'a' && h <= 'f') v |= h - 'a' + 10;
              else if (h >= 'A' && h <= 'F') v |= h - 'A' + 10;
              else return fail();
            }
            p += 4;
            // surrogate pair
            if (v >= 0xD800 && v <= 0xDBFF && p + 6 <= end && p[0] == '\\' &&
                p[1] == 'u') {
              unsigned lo = 0;
              bool okp = true;
              for (int i = 0; i < 4; ++i) {
                char h = p[2 + i];
                lo <<= 4;
                if (h >= '0' && h <= '9') lo |= h - '0';
                else if (h >= 'a' && h <= 'f') lo |= h - 'a' + 10;
                else if (h >= 'A' && h <= 'F') lo |= h - 'A' + 10;
                else { okp = false; break; }
              }
              if (okp && lo >= 0xDC00 && lo <= 0xDFFF) {
                v = 0x10000 + ((v - 0xD800) << 10) + (lo - 0xDC00);
                p += 6;
              }
            }
            // UTF-8 encode
            if (v < 0x80) collect->push_back((char)v);
            else if (v < 0x800) {
              collect->push_back((char)(0xC0 | (v >> 6)));
              collect->push_back((char)(0x80 | (v & 0x3F)));
            } else if (v < 0x10000) {
              collect->push_back((char)(0xE0 | (v >> 12)));
              collect->push_back((char)(0x80 | ((v >> 6) & 0x3F)));
              collect->push_back((char)(0x80 | (v & 0x3F)));
            } else {
              collect->push_back((char)(0xF0 | (v >> 18)));
              collect->push_back((char)(0x80 | ((v >> 12) & 0x3F)));
              collect->push_back((char)(0x80 | ((v >> 6) & 0x3F)));
              collect->push_back((char)(0x80 | (v & 0x3F)));
            }
            break;
          }
          default:
            return fail();
        }
        continue;
      }
    }
    return fail();
  }

  bool fail() {
    ok = false;
    return false;
  }

  bool parse_value(int depth, const std::string& key, bool at_root,
                   bool in_msgs = false) {
    if (depth > 64 || !ok) return fail();
    // chat text is only collected inside the top-level "messages" tree
    // (plus a top-level "system" string) so stray "text"/"content" keys
    // elsewhere in the body never leak into admission/cache-key text —
    // keeps this fast path byte-identical with the strict extractor
    // (aigw/gpu/services.py extract_chat_text)
    bool msgs_root = depth == 1 && key == "messages";
    in_msgs = in_msgs || msgs_root;
    ws();
    if (msgs_root && base != nullptr) msgs_vs = (size_t)(p - base);
    if (p >= end) return fail();
    char c = *p;
    if (c == '{') {
      ++p;
      ws();
      if (p < end && *p == '}') { ++p; return true; }
      while (p < end) {
        std::string k;
        ws();
        if (!parse_string(&k)) return false;
        ws();
        if (p >= end || *p != ':') return fail();
        ++p;
        bool k_is_msgs = depth == 0 && k == "messages";
        if (!parse_value(depth + 1, k, false, in_msgs)) return false;
        if (k_is_msgs && base != nullptr) msgs_ve = (size_t)(p - base);
        ws();
        if (p < end && *p == ',') { ++p; continue; }
        if (p < end && *p == '}') { ++p; return true; }
        return fail();
      }
      return fail();
    }
    if (c == '[') {
      ++p;
      ws();
      if (p < end && *p == ']') { ++p; return true; }
      while (p < end) {
        if (!parse_value(depth + 1, key, false, in_msgs)) return false;
        ws();
        if (p < end && *p == ',') { ++p; continue; }
        if (p < end && *p == ']') { ++p; return true; }
        return fail();
      }
      return fail();
    }
    if (c == '"') {
      bool is_model = depth == 1 && key == "model";
      bool is_text = (in_msgs && (key == "content" || key == "text")) ||
                     (depth == 1 && key == "system");
      if (is_model) {
        // record the value byte span (incl. quotes) when the caller set
        // `base`, so the fast path can splice a model override without
        // re-parsing (sjson-style, translator model override)
        const char* v0 = p;
        bool r = parse_string(&model);
        if (base != nullptr) {
          model_vs = (size_t)(v0 - base);
          model_ve = (size_t)(p - base);
        }
        return r;
      }
      if (is_text) {
        bool r = parse_string(&text);
        text.push_back('\n');
        return r;
      }
      return parse_string(nullptr);
    }
    // literals / numbers
    if (c == 't') {
      if (end - p < 4 || std::memcmp(p, "true", 4) != 0) return fail();
      if (depth == 1 && key == "stream") stream = 1;
      p += 4;
      return true;
    }
    if (c == 'f') {
      if (end - p < 5 || std::memcmp(p, "false", 5) != 0) return fail();
      p += 5;
      return true;
    }
    if (c == 'n') {
      if (end - p < 4 || std::memcmp(p, "null", 4) != 0) return fail();
      p += 4;
      return true;
    }
    // number
    const char* s = p;
    while (p < end && (*p == '-' || *p == '+' || *p == '.' || *p == 'e' ||
                       *p == 'E' || (*p >= '0' && *p <= '9')))
      ++p;
    if (p == s) return fail();
    return true;
  }
};

// Incremental SSE event splitter tolerant of arbitrary chunk boundaries
// (the stateful re-chunking core, reference translator/util.go:40-57).
// Emit is called as emit(event_type, data) for every complete event.
class SSECore {
 public:
  template <class Emit>
  void feed(const char* ptr, size_t n, Emit&& emit) {
    buf_.append(ptr, n);
    size_t start = 0;
    for (;;) {
      size_t nl = buf_.find('\n', start);
      if (nl == std::string::npos) break;
      size_t len = nl - start;
      if (len && buf_[start + len - 1] == '\r') --len;
      feed_line(buf_.data() + start, len, emit);
      start = nl + 1;
    }
    buf_.erase(0, start);
  }

  template <class Emit>
  void flush(Emit&& emit) {
    if (!buf_.empty()) {
      size_t len = buf_.size();
      if (len && buf_[len - 1] == '\r') --len;
      feed_line(buf_.data(), len, emit);
      buf_.clear();
    }
    if (!data_.empty() || !event_.empty()) dispatch(emit);
  }

 private:
  template <class E>
  void feed_line(const char* line, size_t n, E& emit) {
    if (n == 0) {
      if (!data_.empty() || !event_.empty() || has_fields_) dispatch(emit);
      return;
    }
    if (line[0] == ':') return;  // comment
    const char* colon = static_cast<const char*>(memchr(line, ':', n));
    size_t name_len = colon ? (size_t)(colon - line) : n;
    const char* value = colon ? colon + 1 : line + n;
    size_t value_len = colon ? n - name_len - 1 : 0;
    if (value_len && *value == ' ') {
      ++value;
      --value_len;
    }
    if (name_len == 4 && std::memcmp(line, "data", 4) == 0) {
      if (has_data_) data_.push_back('\n');  // joins EMPTY data lines too
      data_.append(value, value_len);
      has_data_ = true;
      has_fields_ = true;
    } else if (name_len == 5 && std::memcmp(line, "event", 5) == 0) {
      event_.assign(value, value_len);
      has_fields_ = true;
    } else if (name_len == 2 && std::memcmp(line, "id", 2) == 0) {
      has_fields_ = true;
    }
  }

  template <class E>
  void dispatch(E& emit) {
    emit(event_, data_);
    data_.clear();
    event_.clear();
    has_fields_ = false;
    has_data_ = false;
  }

  std::string buf_;
  std::string data_;
  std::string event_;
  bool has_fields_ = false;
  bool has_data_ = false;
};

inline bool contains_usage_raw(const char* p, size_t n) {
  return memmem(p, n, "\"usage\"", 7) != nullptr;
}

}  // namespace aigw_core
