// aigw fast path implementation — see fastpath.h for the design notes.
//
// Behavior parity anchors (same invariants as aigw/extproc/server.py and
// the reference):
//   - requests are framed by Content-Length only; any Transfer-Encoding
//     is refused with 501 and conflicting duplicate Content-Length with
//     400 (request-desync hardening, identical to the Python lean front);
//   - client-spoofable x-ai-eg-* / credential-override headers never
//     travel upstream;
//   - model routing happens after the body scan, before backend pick;
//   - per-try translation: every retry splices the ORIGINAL body bytes;
//   - auth (static bearer) is applied over the FINAL body;
//   - local replies (400/404/429/503) are never re-translated;
//   - streamed responses relay the upstream framing unchanged and are
//     tapped for cumulative usage; content-length responses forward
//     byte-identical bodies.

#include "fastpath.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/time.h>
#include <sys/types.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <random>

#include "msgpack_mini.h"
#include "native_core.h"

namespace aigw_fast {

namespace {

int64_t now_ms() {
  return std::chrono::duration_cast<std::chrono::milliseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

int64_t now_us() {
  return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

void set_timeout(int fd, double seconds) {
  struct timeval tv;
  tv.tv_sec = (time_t)seconds;
  tv.tv_usec = (suseconds_t)((seconds - (double)tv.tv_sec) * 1e6);
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
}

void set_nodelay(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

// full-buffer send; false on error/timeout
bool write_all(int fd, const char* p, size_t n) {
  while (n > 0) {
    ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    p += w;
    n -= (size_t)w;
  }
  return true;
}

bool write_all(int fd, const std::string& s) { return write_all(fd, s.data(), s.size()); }

// vectored head+body send: one syscall for the common two-part write
// (saves ~2 syscalls per hop at 80k req/s on a CPU-quota-bound box)
bool write_two(int fd, const std::string& a, const std::string& b) {
  struct iovec iov[2];
  iov[0].iov_base = const_cast<char*>(a.data());
  iov[0].iov_len = a.size();
  iov[1].iov_base = const_cast<char*>(b.data());
  iov[1].iov_len = b.size();
  size_t total = a.size() + b.size();
  size_t sent = 0;
  struct msghdr msg;
  memset(&msg, 0, sizeof(msg));
  msg.msg_iov = iov;
  msg.msg_iovlen = 2;
  while (sent < total) {
    ssize_t w = ::sendmsg(fd, &msg, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    sent += (size_t)w;
    // advance iovecs
    size_t skip = (size_t)w;
    while (msg.msg_iovlen > 0 && skip >= msg.msg_iov[0].iov_len) {
      skip -= msg.msg_iov[0].iov_len;
      ++msg.msg_iov;
      --msg.msg_iovlen;
    }
    if (msg.msg_iovlen > 0) {
      msg.msg_iov[0].iov_base = (char*)msg.msg_iov[0].iov_base + skip;
      msg.msg_iov[0].iov_len -= skip;
    }
  }
  return true;
}

// one recv; 0 = clean EOF, -1 = error/timeout
ssize_t read_some(int fd, char* p, size_t cap) {
  for (;;) {
    ssize_t r = ::recv(fd, p, cap, 0);
    if (r < 0 && errno == EINTR) continue;
    return r;
  }
}

int tcp_connect(const std::string& host, uint16_t port, double timeout_s) {
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_port = htons(port);
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
    struct addrinfo hints, *res = nullptr;
    memset(&hints, 0, sizeof(hints));
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0 || !res) return -1;
    addr.sin_addr = ((struct sockaddr_in*)res->ai_addr)->sin_addr;
    freeaddrinfo(res);
  }
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return -1;
  set_timeout(fd, timeout_s);
  if (::connect(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
    ::close(fd);
    return -1;
  }
  set_nodelay(fd);
  return fd;
}

std::string lower(std::string s) {
  for (auto& c : s) c = (char)tolower((unsigned char)c);
  return s;
}

struct Header {
  std::string name;  // lowercased
  std::string value;
};

struct HttpHead {
  std::string method, path;  // requests
  int status = 0;            // responses
  std::vector<Header> headers;
  size_t head_len = 0;  // bytes consumed incl. CRLFCRLF

  const std::string* get(const char* name) const {
    for (const auto& h : headers)
      if (h.name == name) return &h.value;
    return nullptr;
  }
};

constexpr size_t kMaxHead = 64 * 1024;
constexpr size_t kMaxBody = 64 * 1024 * 1024;

// parse a full head present in buf[0..); returns false on malformed
bool parse_head(const std::string& buf, size_t head_end, bool is_request, HttpHead* out) {
  out->head_len = head_end + 4;
  size_t line_end = buf.find("\r\n");
  if (line_end == std::string::npos || line_end > head_end) return false;
  const std::string first = buf.substr(0, line_end);
  if (is_request) {
    size_t sp1 = first.find(' ');
    if (sp1 == std::string::npos) return false;
    size_t sp2 = first.find(' ', sp1 + 1);
    if (sp2 == std::string::npos) return false;
    out->method = first.substr(0, sp1);
    out->path = first.substr(sp1 + 1, sp2 - sp1 - 1);
  } else {
    if (first.compare(0, 7, "HTTP/1.") != 0) return false;
    size_t sp = first.find(' ');
    if (sp == std::string::npos) return false;
    out->status = atoi(first.c_str() + sp + 1);
  }
  size_t pos = line_end + 2;
  while (pos < head_end) {
    size_t eol = buf.find("\r\n", pos);
    if (eol == std::string::npos || eol > head_end) eol = head_end;
    size_t colon = buf.find(':', pos);
    if (colon != std::string::npos && colon < eol) {
      Header h;
      h.name = lower(buf.substr(pos, colon - pos));
      // trim
      size_t vs = colon + 1;
      while (vs < eol && (buf[vs] == ' ' || buf[vs] == '\t')) ++vs;
      size_t ve = eol;
      while (ve > vs && (buf[ve - 1] == ' ' || buf[ve - 1] == '\t')) --ve;
      while (!h.name.empty() && (h.name.back() == ' ' || h.name.back() == '\t'))
        h.name.pop_back();
      h.value = buf.substr(vs, ve - vs);
      out->headers.push_back(std::move(h));
    }
    pos = eol + 2;
  }
  return true;
}

// -------- body framing ------------------------------------------------------

enum class Framing { kLength, kChunked, kClose, kNone };

Framing response_framing(const HttpHead& h, int64_t* length) {
  const std::string* te = h.get("transfer-encoding");
  if (te && lower(*te).find("chunked") != std::string::npos) return Framing::kChunked;
  if (h.status == 204 || h.status == 304 || (h.status >= 100 && h.status < 200)) {
    *length = 0;
    return Framing::kLength;
  }
  const std::string* cl = h.get("content-length");
  if (cl) {
    *length = atoll(cl->c_str());
    return Framing::kLength;
  }
  return Framing::kClose;
}

// incremental chunked-transfer parser: feeds raw bytes, emits payload
// spans, reports completion. Used both to find the response end for
// connection reuse and to tap SSE payloads mid-relay.
class ChunkedParser {
 public:
  bool done() const { return state_ == State::kDone; }
  bool error() const { return state_ == State::kError; }

  // feed raw bytes; calls payload(p, n) for chunk-data spans
  template <class F>
  void feed(const char* p, size_t n, F&& payload) {
    size_t i = 0;
    while (i < n && state_ != State::kDone && state_ != State::kError) {
      switch (state_) {
        case State::kSize: {
          char c = p[i++];
          line_.push_back(c);
          if (line_.size() > 256) { state_ = State::kError; break; }
          if (c == '\n') {
            size_t semi = line_.find(';');
            std::string hex = line_.substr(0, semi);
            remaining_ = strtoll(hex.c_str(), nullptr, 16);
            line_.clear();
            if (remaining_ < 0) state_ = State::kError;
            else if (remaining_ == 0) state_ = State::kTrailer;
            else state_ = State::kData;
          }
          break;
        }
        case State::kData: {
          size_t take = std::min((size_t)remaining_, n - i);
          payload(p + i, take);
          i += take;
          remaining_ -= (int64_t)take;
          if (remaining_ == 0) state_ = State::kDataCrlf, crlf_ = 0;
          break;
        }
        case State::kDataCrlf: {
          char c = p[i++];
          if (c == '\n') state_ = State::kSize;
          else if (c != '\r') state_ = State::kError;
          break;
        }
        case State::kTrailer: {
          char c = p[i++];
          line_.push_back(c);
          if (line_.size() > 4096) { state_ = State::kError; break; }
          if (c == '\n') {
            if (line_ == "\r\n" || line_ == "\n") state_ = State::kDone;
            line_.clear();
          }
          break;
        }
        default:
          break;
      }
    }
  }

 private:
  enum class State { kSize, kData, kDataCrlf, kTrailer, kDone, kError };
  State state_ = State::kSize;
  std::string line_;
  int64_t remaining_ = 0;
  int crlf_ = 0;
};

// -------- usage extraction --------------------------------------------------

// scan a JSON fragment for the token counters inside its "usage" object
// (translator/openai_openai.go:185-223 equivalent: cumulative counts, the
// caller keeps the max). Flat byte scan — the fragment is trusted JSON
// from the upstream and the keys are unambiguous in OpenAI responses.
bool extract_usage(const char* p, size_t n, Usage* u) {
  const char* up = (const char*)memmem(p, n, "\"usage\"", 7);
  if (!up) return false;
  size_t off = (size_t)(up - p);
  auto grab = [&](const char* key, size_t klen) -> int64_t {
    const char* k = (const char*)memmem(p + off, n - off, key, klen);
    if (!k) return -1;
    const char* q = k + klen;
    const char* e = p + n;
    while (q < e && (*q == ':' || *q == ' ' || *q == '\t')) ++q;
    if (q >= e || *q < '0' || *q > '9') return -1;
    int64_t v = 0;
    while (q < e && *q >= '0' && *q <= '9') v = v * 10 + (*q++ - '0');
    return v;
  };
  int64_t in = grab("\"prompt_tokens\"", 15);
  int64_t out = grab("\"completion_tokens\"", 19);
  int64_t total = grab("\"total_tokens\"", 14);
  if (in < 0) in = grab("\"input_tokens\"", 14);
  if (out < 0) out = grab("\"output_tokens\"", 15);
  bool any = false;
  if (in > 0) u->input = std::max(u->input, in), any = true;
  if (out > 0) u->output = std::max(u->output, out), any = true;
  if (total > 0) u->total = std::max(u->total, total), any = true;
  if (u->total == 0 && (u->input || u->output)) u->total = u->input + u->output;
  return any;
}

// -------- body splices ------------------------------------------------------

// replace the top-level "model" value span with the override
std::string splice_model(const std::string& body, size_t vs, size_t ve,
                         const std::string& override_model) {
  std::string out;
  out.reserve(body.size() + override_model.size() + 2);
  out.append(body, 0, vs);
  out.push_back('"');
  out += override_model;  // config-supplied, no escaping needed
  out.push_back('"');
  out.append(body, ve, body.size() - ve);
  return out;
}

// force stream_options.include_usage=true (endpointspec.go:138-154): when
// the body has no stream_options, insert one before the final '}'.
bool splice_include_usage(std::string* body) {
  if (memmem(body->data(), body->size(), "\"stream_options\"", 16)) return false;
  size_t close = body->rfind('}');
  if (close == std::string::npos) return false;
  body->insert(close, ",\"stream_options\":{\"include_usage\":true}");
  return true;
}

}  // namespace

// -------- upstream pool -----------------------------------------------------

class UpstreamPool {
 public:
  int acquire(const std::string& host, uint16_t port, double timeout_s) {
    const std::string key = host + ":" + std::to_string(port);
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = idle_.find(key);
      while (it != idle_.end() && !it->second.empty()) {
        int fd = it->second.back();
        it->second.pop_back();
        // liveness probe: a reusable idle conn has no readable bytes;
        // readable or EOF means the peer closed or sent garbage
        char tmp;
        ssize_t r = ::recv(fd, &tmp, 1, MSG_PEEK | MSG_DONTWAIT);
        if (r == -1 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
          set_timeout(fd, timeout_s);
          return fd;
        }
        ::close(fd);
      }
    }
    return tcp_connect(host, port, timeout_s);
  }

  void release(const std::string& host, uint16_t port, int fd, bool reusable) {
    if (!reusable) {
      ::close(fd);
      return;
    }
    const std::string key = host + ":" + std::to_string(port);
    std::lock_guard<std::mutex> lk(mu_);
    auto& v = idle_[key];
    if (v.size() >= 512) {
      ::close(fd);
      return;
    }
    v.push_back(fd);
  }

  void close_all() {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& kv : idle_)
      for (int fd : kv.second) ::close(fd);
    idle_.clear();
  }

 private:
  std::mutex mu_;
  std::map<std::string, std::vector<int>> idle_;
};

// -------- GPU admission client ----------------------------------------------

class GpuAdmissionClient {
 public:
  struct Waiter {
    std::mutex m;
    std::condition_variable cv;
    bool done = false;
    int64_t count = 0;
  };

  bool start(const std::string& socket_path, int window_us, int max_batch) {
    window_us_ = window_us;
    max_batch_ = max_batch;
    fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd_ < 0) return false;
    struct sockaddr_un addr;
    memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    strncpy(addr.sun_path, socket_path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(fd_, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
      ::close(fd_);
      fd_ = -1;
      return false;
    }
    batcher_ = std::thread([this] { batch_loop(); });
    reader_ = std::thread([this] { read_loop(); });
    return true;
  }

  void stop() {
    stopping_ = true;
    {
      std::lock_guard<std::mutex> lk(qmu_);
      qcv_.notify_all();
    }
    if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
    if (batcher_.joinable()) batcher_.join();
    if (reader_.joinable()) reader_.join();
    if (fd_ >= 0) ::close(fd_), fd_ = -1;
    fail_all();
  }

  ~GpuAdmissionClient() { stop(); }

  // blocks until the batched RPC answers; 0 on service failure (the
  // request proceeds without GPU accounting, like the Python path when
  // gpu_services is absent)
  int64_t count_text(const std::string& text) {
    if (stopping_ || fd_ < 0) return 0;
    auto w = std::make_shared<Waiter>();
    {
      std::lock_guard<std::mutex> lk(qmu_);
      q_texts_.push_back(text);
      q_waiters_.push_back(w);
      qcv_.notify_one();
    }
    std::unique_lock<std::mutex> lk(w->m);
    w->cv.wait_for(lk, std::chrono::seconds(30), [&] { return w->done; });
    return w->count;
  }

 private:
  void batch_loop() {
    while (!stopping_) {
      std::vector<std::string> texts;
      std::vector<std::shared_ptr<Waiter>> waiters;
      {
        std::unique_lock<std::mutex> lk(qmu_);
        qcv_.wait(lk, [&] { return stopping_ || !q_texts_.empty(); });
        if (stopping_) return;
      }
      // micro-batch window: collect arrivals across ALL connections
      if (window_us_ > 0)
        std::this_thread::sleep_for(std::chrono::microseconds(window_us_));
      {
        std::lock_guard<std::mutex> lk(qmu_);
        size_t take = std::min(q_texts_.size(), (size_t)max_batch_);
        texts.assign(q_texts_.begin(), q_texts_.begin() + take);
        waiters.assign(q_waiters_.begin(), q_waiters_.begin() + take);
        q_texts_.erase(q_texts_.begin(), q_texts_.begin() + take);
        q_waiters_.erase(q_waiters_.begin(), q_waiters_.begin() + take);
      }
      if (texts.empty()) continue;
      uint64_t id = next_id_++;
      {
        std::lock_guard<std::mutex> lk(pmu_);
        pending_[id] = waiters;
      }
      MsgpackWriter w;
      w.map_header(3);
      w.str("id", 2);
      w.uint(id);
      w.str("op", 2);
      w.str("count_batch", 11);
      w.str("texts", 5);
      w.array_header((uint32_t)texts.size());
      for (const auto& t : texts) w.bin(t.data(), t.size());
      uint32_t len = (uint32_t)w.out.size();
      char hdr[4] = {(char)(len & 0xff), (char)((len >> 8) & 0xff),
                     (char)((len >> 16) & 0xff), (char)((len >> 24) & 0xff)};
      std::string frame(hdr, 4);
      frame += w.out;
      if (!write_all(fd_, frame)) {
        stopping_ = true;
        fail_all();
        return;
      }
    }
  }

  void read_loop() {
    std::string buf;
    char tmp[65536];
    while (!stopping_) {
      ssize_t r = read_some(fd_, tmp, sizeof(tmp));
      if (r <= 0) break;
      buf.append(tmp, (size_t)r);
      while (buf.size() >= 4) {
        uint32_t len = (uint8_t)buf[0] | ((uint8_t)buf[1] << 8) |
                       ((uint8_t)buf[2] << 16) | ((uint8_t)buf[3] << 24);
        if (buf.size() < 4 + (size_t)len) break;
        handle_reply(buf.data() + 4, len);
        buf.erase(0, 4 + (size_t)len);
      }
    }
    stopping_ = true;
    fail_all();
  }

  void handle_reply(const char* p, size_t n) {
    MsgpackReader r(p, n);
    int pairs = r.map_header();
    uint64_t id = 0;
    std::vector<int64_t> counts;
    for (int i = 0; i < pairs && r.ok(); ++i) {
      std::string key;
      if (!r.str(&key)) return;
      if (key == "id") {
        int64_t v;
        if (!r.integer(&v)) return;
        id = (uint64_t)v;
      } else if (key == "counts") {
        int m = r.array_header();
        for (int j = 0; j < m && r.ok(); ++j) {
          int64_t v;
          if (!r.integer(&v)) return;
          counts.push_back(v);
        }
      } else {
        r.skip();
      }
    }
    std::vector<std::shared_ptr<Waiter>> waiters;
    {
      std::lock_guard<std::mutex> lk(pmu_);
      auto it = pending_.find(id);
      if (it == pending_.end()) return;
      waiters = std::move(it->second);
      pending_.erase(it);
    }
    for (size_t i = 0; i < waiters.size(); ++i) {
      auto& w = waiters[i];
      std::lock_guard<std::mutex> lk(w->m);
      w->count = i < counts.size() ? counts[i] : 0;
      w->done = true;
      w->cv.notify_all();
    }
  }

  void fail_all() {
    std::vector<std::shared_ptr<Waiter>> all;
    {
      std::lock_guard<std::mutex> lk(qmu_);
      all.insert(all.end(), q_waiters_.begin(), q_waiters_.end());
      q_waiters_.clear();
      q_texts_.clear();
    }
    {
      std::lock_guard<std::mutex> lk(pmu_);
      for (auto& kv : pending_)
        all.insert(all.end(), kv.second.begin(), kv.second.end());
      pending_.clear();
    }
    for (auto& w : all) {
      std::lock_guard<std::mutex> lk(w->m);
      w->done = true;
      w->cv.notify_all();
    }
  }

  int fd_ = -1;
  int window_us_ = 100;
  int max_batch_ = 256;
  std::atomic<bool> stopping_{false};
  std::atomic<uint64_t> next_id_{1};
  std::mutex qmu_;
  std::condition_variable qcv_;
  std::vector<std::string> q_texts_;
  std::vector<std::shared_ptr<GpuAdmissionClient::Waiter>> q_waiters_;
  std::mutex pmu_;
  std::map<uint64_t, std::vector<std::shared_ptr<Waiter>>> pending_;
  std::thread batcher_, reader_;
};


// -------- in-process HIP admission batcher ----------------------------------

// admission.hip entry points (same .so, separate HIP translation unit)
GpuAdmissionDirect* admission_create(const long long* htab_keys,
                                     const int32_t* htab_rank, int htab_n,
                                     size_t max_bytes, int max_req, int device);
bool admission_count(GpuAdmissionDirect* a, const char* bytes, size_t n,
                     const int64_t* offsets, int n_req, int32_t* counts_out);
void admission_destroy(GpuAdmissionDirect* a);
bool admission_init_cache(GpuAdmissionDirect* a, const uint16_t* emb, int vocab,
                          const uint16_t* proj, int dim, long long capacity,
                          float threshold, int pending_cap, bool fp8);
bool admission_count_lookup(GpuAdmissionDirect* a, const char* bytes, size_t n,
                            const int64_t* offsets, int n_req,
                            int32_t* counts_out, const int32_t* pending_slots,
                            int32_t* rows_out, float* scores_out);
long long admission_cache_insert(GpuAdmissionDirect* a, int pending_slot);
bool admission_submit(GpuAdmissionDirect* a, int set, const char* bytes,
                      size_t n, const int64_t* offsets, int n_req,
                      const int32_t* pending_slots);
bool admission_wait(GpuAdmissionDirect* a, int set, int n_req,
                    int32_t* counts_out, int32_t* rows_out, float* scores_out);
char* admission_staging(GpuAdmissionDirect* a, int set);

// Adaptive batching with NO timer window: one batcher thread drains
// whatever accumulated while the previous GPU batch ran — the kernel
// duration itself is the coalescing window, so light load gets ~zero
// added latency and heavy load gets large batches automatically.
class DirectGpuBatcher {
 public:
  bool start(const long long* htab_keys, const int32_t* htab_rank, int htab_n,
             int max_batch, size_t max_bytes, int max_req, int device) {
    adm_ = admission_create(htab_keys, htab_rank, htab_n, max_bytes, max_req,
                            device);
    if (adm_ == nullptr) return false;
    max_batch_ = std::min(max_batch, max_req);
    max_bytes_ = max_bytes;
    worker_ = std::thread([this] { loop(); });
    return true;
  }

  void stop() {
    stopping_ = true;
    {
      std::lock_guard<std::mutex> lk(mu_);
      cv_.notify_all();
    }
    if (worker_.joinable()) worker_.join();
    if (adm_ != nullptr) admission_destroy(adm_), adm_ = nullptr;
    fail_all();
  }

  ~DirectGpuBatcher() { stop(); }

  int64_t count_text(const std::string& text) {
    CacheLookup cl = count_lookup_text(text);
    // the batcher parks a query vector whenever the cache is on; a
    // count-only caller (streamed requests) must return the slot or the
    // 4096-slot pending pool starves after enough streams
    cache_release_slot(cl.slot);
    return cl.tokens;
  }

  struct Waiter2 {
    std::mutex m;
    std::condition_variable cv;
    bool done = false;
    int64_t count = 0;
    int32_t row = -1;
    float score = 0.f;
    int32_t slot = -1;
  };

  struct CacheLookup {
    int64_t tokens = 0;
    int32_t row = -1;      // index row of a hit (score >= threshold)
    float score = 0.f;
    int32_t slot = -1;     // pending-pool slot holding the query vector
  };

  // Async split: enqueue() submits a text for the next batch and
  // returns a handle; collect() blocks until the batch completes. The
  // serving path enqueues BEFORE dispatching upstream and collects at
  // response time, hiding the admission batch behind the upstream
  // round-trip (the reference also charges usage post-response:
  // processor_impl.go buildDynamicMetadata runs on response complete).
  std::shared_ptr<Waiter2> enqueue(const std::string& text) {
    if (stopping_) return nullptr;
    auto w = std::make_shared<Waiter2>();
    {
      std::lock_guard<std::mutex> lk(mu_);
      q_texts_.push_back(text);
      q_waiters_.push_back(w);
      cv_.notify_one();
    }
    return w;
  }

  CacheLookup collect(const std::shared_ptr<Waiter2>& w) {
    CacheLookup out;
    if (w == nullptr) return out;
    std::unique_lock<std::mutex> lk(w->m);
    w->cv.wait_for(lk, std::chrono::seconds(30), [&] { return w->done; });
    out.tokens = w->count;
    out.row = w->row;
    out.score = w->score;
    out.slot = w->slot;
    return out;
  }

  // count + (when the cache is enabled) lookup in ONE GPU batch. The
  // returned slot (if >= 0) must be passed to cache_insert_slot() or
  // cache_release_slot() by the caller exactly once.
  CacheLookup count_lookup_text(const std::string& text) {
    return collect(enqueue(text));
  }

  bool init_cache(const uint16_t* emb, int vocab, const uint16_t* proj,
                  int dim, long long capacity, float threshold, bool fp8) {
    if (adm_ == nullptr) return false;
    const int pending_cap = 4096;
    if (!admission_init_cache(adm_, emb, vocab, proj, dim, capacity, threshold,
                              pending_cap, fp8))
      return false;
    {
      std::lock_guard<std::mutex> lk(slot_mu_);
      free_slots_.resize(pending_cap);
      for (int i = 0; i < pending_cap; ++i) free_slots_[i] = pending_cap - 1 - i;
    }
    cache_on_ = true;
    return true;
  }

  bool cache_on() const { return cache_on_; }

  // append the parked query vector to the index; frees the slot. The
  // index-ring state is guarded by insert_mu_ (callers also hold the
  // server's value-store lock for the row->value write).
  long long cache_insert_slot(int slot) {
    long long row;
    {
      std::lock_guard<std::mutex> lk(insert_mu_);
      row = admission_cache_insert(adm_, slot);
    }
    cache_release_slot(slot);
    return row;
  }

  void cache_release_slot(int slot) {
    if (slot < 0) return;
    std::lock_guard<std::mutex> lk(slot_mu_);
    free_slots_.push_back(slot);
  }


 private:
  // Two-deep software pipeline over the two admission BatchSets: while
  // batch N runs on set s's stream, batch N+1 is collected, packed, and
  // submitted on set 1-s. The GPU pipeline is launch/latency-bound per
  // batch (~1.5 ms), so overlapping whole batches (plus the CPU-side
  // pack/fulfill work) nearly doubles admission throughput under load.
  struct InFlight {
    std::vector<std::shared_ptr<Waiter2>> waiters;
    std::vector<int32_t> slots;
    int64_t t0 = 0;
    bool valid = false;
  };

  // retire a submitted batch: block on its completion event, decode the
  // D2H results, fulfill the waiters, record batch stats
  void finish_set(int s) {
    InFlight& fl = inflight_[s];
    if (!fl.valid) return;
    int n = (int)fl.waiters.size();
    std::vector<int32_t> counts((size_t)n, 0);
    std::vector<int32_t> rows((size_t)n, -1);
    std::vector<float> scores((size_t)n, 0.f);
    bool want_cache = cache_on_.load();
    int64_t w0 = now_us();
    if (!admission_wait(adm_, s, n, counts.data(),
                        want_cache ? rows.data() : nullptr,
                        want_cache ? scores.data() : nullptr))
      stats_errors++;
    int64_t w1 = now_us();
    stats_wait_us += (uint64_t)(w1 - w0);
    int64_t bt = w1 - fl.t0;
    stats_batches++;
    stats_texts += (uint64_t)n;
    stats_time_us += (uint64_t)bt;
    uint64_t prev = stats_max_us.load();
    while ((uint64_t)bt > prev &&
           !stats_max_us.compare_exchange_weak(prev, (uint64_t)bt)) {}
    for (int i = 0; i < n; ++i) {
      auto& w = fl.waiters[i];
      std::lock_guard<std::mutex> lk(w->m);
      w->count = counts[i];
      w->row = rows[i];
      w->score = scores[i];
      w->slot = fl.slots[i];
      w->done = true;
      w->cv.notify_all();
    }
    stats_fulfill_us += (uint64_t)(now_us() - w1);
    fl.waiters.clear();
    fl.slots.clear();
    fl.valid = false;
  }

  void loop() {
    std::vector<int64_t> offs;
    int next_set = 0;
    while (!stopping_) {
      std::vector<std::string> texts;
      std::vector<std::shared_ptr<Waiter2>> waiters;
      {
        std::unique_lock<std::mutex> lk(mu_);
        if (q_texts_.empty()) {
          int other = 1 - next_set;
          if (inflight_[other].valid) {
            // nothing queued yet: retire the running batch first (its
            // kernel time is the coalescing window for the next batch)
            lk.unlock();
            finish_set(other);
            continue;
          }
          cv_.wait(lk, [&] { return stopping_ || !q_texts_.empty(); });
          if (stopping_) break;
        }
        size_t take = 0, bytes = 0;
        while (take < q_texts_.size() && (int)take < max_batch_ &&
               bytes + q_texts_[take].size() <= max_bytes_)
          bytes += q_texts_[take++].size();
        if (take == 0) take = 1;  // oversized single text: truncated below
        texts.assign(q_texts_.begin(), q_texts_.begin() + take);
        waiters.assign(q_waiters_.begin(), q_waiters_.begin() + take);
        q_texts_.erase(q_texts_.begin(), q_texts_.begin() + take);
        q_waiters_.erase(q_waiters_.begin(), q_waiters_.begin() + take);
      }
      finish_set(next_set);  // the set must be idle before its staging reuse
      int64_t p0 = now_us();
      // pack straight into the idle set's pinned staging buffer: the
      // copy into `packed` + submit's memcpy into staging was two
      // passes over the same ~2.5 MB (stats_submit_us showed the
      // staging memcpy as most of the submit phase)
      char* stage = admission_staging(adm_, next_set);
      size_t used = 0;
      offs.clear();
      for (auto& t : texts) {
        offs.push_back((int64_t)used);
        size_t room = max_bytes_ - used;
        size_t take_n = std::min(t.size(), room);
        memcpy(stage + used, t.data(), take_n);
        used += take_n;
      }
      size_t nt = texts.size();
      std::vector<int32_t> slots((size_t)nt, -1);
      if (cache_on_) {
        std::lock_guard<std::mutex> lk(slot_mu_);
        for (size_t i = 0; i < nt && !free_slots_.empty(); ++i) {
          slots[i] = free_slots_.back();
          free_slots_.pop_back();
        }
      }
      stats_pack_us += (uint64_t)(now_us() - p0);
      int64_t s0 = now_us();
      bool ok = used > 0 &&
                admission_submit(adm_, next_set, nullptr, used,
                                 offs.data(), (int)nt,
                                 cache_on_ ? slots.data() : nullptr);
      stats_submit_us += (uint64_t)(now_us() - s0);
      if (!ok) {
        if (used > 0) stats_errors++;
        for (size_t i = 0; i < nt; ++i) {
          auto& w = waiters[i];
          std::lock_guard<std::mutex> lk(w->m);
          w->slot = slots[i];
          w->done = true;
          w->cv.notify_all();
        }
        continue;
      }
      InFlight& fl = inflight_[next_set];
      fl.waiters = std::move(waiters);
      fl.slots = std::move(slots);
      fl.t0 = now_us();
      fl.valid = true;
      // overlap: retire the OLDER batch while this one runs on the GPU
      finish_set(1 - next_set);
      next_set ^= 1;
    }
    finish_set(0);
    finish_set(1);
  }

  void fail_all() {
    std::vector<std::shared_ptr<Waiter2>> all;
    {
      std::lock_guard<std::mutex> lk(mu_);
      all = std::move(q_waiters_);
      q_waiters_.clear();
      q_texts_.clear();
    }
    for (auto& w : all) {
      std::lock_guard<std::mutex> lk(w->m);
      w->done = true;
      w->cv.notify_all();
    }
  }

  GpuAdmissionDirect* adm_ = nullptr;
  std::atomic<bool> cache_on_{false};
  std::mutex slot_mu_;
  std::vector<int32_t> free_slots_;
  std::mutex insert_mu_;

 public:
  std::atomic<uint64_t> stats_batches{0};
  std::atomic<uint64_t> stats_texts{0};
  std::atomic<uint64_t> stats_time_us{0};
  std::atomic<uint64_t> stats_max_us{0};
  std::atomic<uint64_t> stats_errors{0};
  // phase breakdown (batcher thread wall time per phase, summed)
  std::atomic<uint64_t> stats_pack_us{0};
  std::atomic<uint64_t> stats_submit_us{0};
  std::atomic<uint64_t> stats_wait_us{0};
  std::atomic<uint64_t> stats_fulfill_us{0};

 private:
  int max_batch_ = 1024;
  size_t max_bytes_ = 0;
  std::atomic<bool> stopping_{false};
  std::mutex mu_;
  std::condition_variable cv_;
  std::vector<std::string> q_texts_;
  std::vector<std::shared_ptr<Waiter2>> q_waiters_;
  InFlight inflight_[2];
  std::thread worker_;
};

// -------- connection handler ------------------------------------------------


namespace {

// client headers never forwarded upstream (aigw/extproc/server.py
// _HOP_BY_HOP + internal x-ai-eg-* spoof-strip + override-strip)
bool banned_upstream_header(const std::string& name) {
  static const char* kBanned[] = {
      "host", "content-length", "content-type", "connection", "keep-alive",
      "proxy-authenticate", "proxy-authorization", "te", "trailer",
      "transfer-encoding", "upgrade", "expect", "accept-encoding",
      "authorization"};
  for (const char* b : kBanned)
    if (name == b) return true;
  if (name.compare(0, 8, "x-ai-eg-") == 0) return true;
  if (name.compare(0, 7, "x-aigw-") == 0) return true;
  return false;
}

}  // namespace

class ConnHandler {
 public:
  ConnHandler(FastServer* srv, int fd) : srv_(srv), fd_(fd) {}

  void run() {
    std::string buf;
    char tmp[65536];
    for (;;) {
      // locate a complete head
      size_t head_end;
      for (;;) {
        head_end = buf.find("\r\n\r\n");
        if (head_end != std::string::npos) break;
        if (buf.size() > kMaxHead) return;
        ssize_t r = read_some(fd_, tmp, sizeof(tmp));
        if (r <= 0) return;
        srv_->stats_.bytes_in += (uint64_t)r;
        buf.append(tmp, (size_t)r);
      }
      HttpHead req;
      if (!parse_head(buf, head_end, true, &req)) {
        simple_reply(400, "invalid_request_error", "malformed request", true);
        return;
      }
      // desync hardening (identical to the Python lean front)
      int64_t clen = 0;
      bool te = false, dup_conflict = false;
      {
        const std::string* seen = nullptr;
        for (const auto& h : req.headers) {
          if (h.name == "transfer-encoding") te = true;
          if (h.name == "content-length") {
            if (seen && *seen != h.value) dup_conflict = true;
            seen = &h.value;
          }
        }
        if (seen) clen = atoll(seen->c_str());
      }
      if (te) {
        simple_reply(501, "invalid_request_error", "transfer-encoding not supported", true);
        return;
      }
      if (dup_conflict || clen < 0 || (size_t)clen > kMaxBody) {
        simple_reply(400, "invalid_request_error", "bad content-length", true);
        return;
      }
      while (buf.size() < req.head_len + (size_t)clen) {
        ssize_t r = read_some(fd_, tmp, sizeof(tmp));
        if (r <= 0) return;
        srv_->stats_.bytes_in += (uint64_t)r;
        buf.append(tmp, (size_t)r);
      }
      std::string body = buf.substr(req.head_len, (size_t)clen);
      buf.erase(0, req.head_len + (size_t)clen);

      bool keep = handle_request(req, body);
      if (!keep) return;
    }
  }

 private:
  bool handle_request(const HttpHead& req, const std::string& body) {
    srv_->stats_.requests++;
    if (req.method == "GET" && req.path == "/health") {
      return raw_reply(200, "application/json",
                       "{\"status\":\"ok\",\"front\":\"fast\"}");
    }
    bool hot = req.method == "POST" &&
               (req.path == "/v1/chat/completions" || req.path == "/v1/completions" ||
                req.path == "/v1/embeddings");
    if (!hot) return fallback(req, body);

    int64_t t0 = now_us();
    // one-pass native scan: model + stream + chat text + model span
    aigw_core::Scan sc{body.data(), body.data() + body.size()};
    sc.base = body.data();
    bool ok = sc.parse_value(0, "", true) && sc.ok;
    if (ok) {
      sc.ws();
      ok = sc.p == sc.end;
    }
    if (!ok) {
      srv_->stats_.responses_4xx++;
      return simple_reply(400, "invalid_request_error",
                          "invalid request body: malformed JSON", false);
    }
    if (sc.model.empty()) {
      srv_->stats_.responses_4xx++;
      return simple_reply(400, "invalid_request_error",
                          "missing required field 'model'", false);
    }
    // snapshot the route table: a concurrent swap_routes (hot reload)
    // cannot free it under us, and this request finishes on the table
    // it resolved (watcher.go:79-160 swap semantics)
    std::shared_ptr<const std::vector<FastRoute>> table =
        std::atomic_load(&srv_->routes_);
    const FastRoute* route = nullptr;
    for (const auto& r : *table) {
      if (r.catch_all || r.model_match == sc.model) {
        route = &r;
        break;
      }
    }
    if (!route) {
      srv_->stats_.responses_4xx++;
      return simple_reply(404, "model_not_found",
                          "no route matched model " + sc.model, false);
    }
    if (!route->eligible) return fallback(req, body);
    bool stream = sc.stream == 1 && req.path != "/v1/embeddings";

    double retry_after = 0;
    std::string rule_name;
    if (!srv_->rl_check(&retry_after, &rule_name)) {
      srv_->stats_.local_429++;
      srv_->stats_.responses_4xx++;
      std::string extra = "retry-after: " + std::to_string((int)retry_after + 1) + "\r\n";
      return simple_reply(429, "rate_limit_exceeded", "token budget exhausted",
                          false, extra);
    }

    int64_t gpu_tokens = 0;
    int32_t cache_slot = -1;
    uint64_t cache_fp = 0;
    pending_count_.reset();  // per-request deferred admission handle
    bool cache_eligible = srv_->gpu_direct_ != nullptr &&
                          srv_->gpu_direct_->cache_on() && !stream &&
                          req.path == "/v1/chat/completions" &&
                          !sc.text.empty();
    if (cache_eligible) {
      // scope fingerprint: near-identical prompts may embed to the same
      // vector, so the VALUE carries a hash of everything that must not
      // be shared — model, route, client credential, and every body byte
      // OUTSIDE the messages array (sampling params etc.); compared at
      // hit time (same rule as the Python cache's tag)
      cache_fp = 1469598103934665603ull;  // FNV-1a 64
      auto fnv = [&](const char* p, size_t n) {
        for (size_t i = 0; i < n; ++i) {
          cache_fp ^= (unsigned char)p[i];
          cache_fp *= 1099511628211ull;
        }
      };
      fnv(route->name.data(), route->name.size());
      fnv(sc.model.data(), sc.model.size());
      const std::string* auth_h = req.get("authorization");
      if (auth_h) fnv(auth_h->data(), auth_h->size());
      if (sc.msgs_ve > sc.msgs_vs && sc.msgs_ve <= body.size()) {
        fnv(body.data(), sc.msgs_vs);
        fnv(body.data() + sc.msgs_ve, body.size() - sc.msgs_ve);
      } else {
        fnv(body.data(), body.size());
      }
      auto cl = srv_->gpu_direct_->count_lookup_text(sc.text);
      gpu_tokens = cl.tokens;
      srv_->stats_.gpu_tokens += (uint64_t)gpu_tokens;
      cache_slot = cl.slot;
      if (cl.row >= 0) {
        std::string hit;
        {
          std::lock_guard<std::mutex> lk(srv_->cache_mu_);
          if ((size_t)cl.row < srv_->cache_values_.size())
            hit = srv_->cache_values_[cl.row];
        }
        if (hit.size() > 9 &&
            memcmp(hit.data(), &cache_fp, 8) == 0 && hit[8] == 'U') {
          srv_->gpu_direct_->cache_release_slot(cache_slot);
          srv_->stats_.cache_hits++;
          srv_->stats_.responses_2xx++;
          record_latency(t0);
          std::string h =
              "HTTP/1.1 200 X\r\ncontent-type: application/json\r\n"
              "x-aigw-cache: hit\r\ncontent-length: " +
              std::to_string(hit.size() - 9) + "\r\n\r\n";
          std::string bodyv = hit.substr(9);
          srv_->stats_.bytes_out += h.size() + bodyv.size();
          return write_two(fd_, h, bodyv);
        }
      }
      srv_->stats_.cache_misses++;
    } else if (srv_->gpu_enabled() && req.path == "/v1/chat/completions" &&
               !sc.text.empty()) {
      if (srv_->gpu_direct_ != nullptr && stream) {
        // STREAMS overlap the admission batch with the upstream
        // round-trip: the count is only consumed at stream end (usage
        // fallback + rate-limit charge), so enqueue now and collect in
        // finish_usage — measured 112.6k -> 124.4k req/s. Unary keeps
        // the synchronous wait: deferring it measured 116.7k -> 107.4k
        // @ p99 2.1 ms, because the ~1 ms admission sleep paces the
        // 128 relay threads under the CPU quota (without it they all
        // stay runnable and earn CFS throttling).
        pending_count_ = srv_->gpu_direct_->enqueue(sc.text);
      } else {
        gpu_tokens = srv_->gpu_count(sc.text);
        srv_->stats_.gpu_tokens += (uint64_t)gpu_tokens;
      }
    }

    // attempt order: tiers by priority, weighted shuffle inside a tier
    // (single-backend routes — the common case — skip the machinery)
    std::vector<const FastBackend*> order;
    if (route->backends.size() == 1)
      order.push_back(&route->backends[0]);
    else
      order = attempt_order(*route);
    int attempts_left = std::min((int)order.size(), route->retries + 1);
    bool first = true;
    for (const FastBackend* be : order) {
      if (attempts_left-- <= 0) break;
      if (!first) srv_->stats_.retries++;
      first = false;
      int outcome = try_backend(req, body, sc, *route, *be, stream, gpu_tokens,
                                t0, cache_slot, cache_fp);
      if (outcome == 0) return true;    // handled, keep-alive
      if (outcome == 2) return false;   // handled, close
      // outcome 1: retriable failure — next backend re-splices the
      // ORIGINAL body (per-try translation, A.8)
    }
    if (cache_slot >= 0) srv_->gpu_direct_->cache_release_slot(cache_slot);
    pending_count_.reset();  // failed request: nothing to charge
    srv_->stats_.responses_5xx++;
    return simple_reply(503, "upstream_error", "no healthy upstream", false);
  }

  std::vector<const FastBackend*> attempt_order(const FastRoute& route) {
    std::vector<const FastBackend*> order;
    order.reserve(route.backends.size());
    // backends are pre-sorted by priority; weighted order inside a tier
    // (Efraimidis-Spirakis keys, matching aigw/extproc/router.py)
    static thread_local std::mt19937_64 rng{std::random_device{}()};
    std::uniform_real_distribution<double> uni(1e-12, 1.0);
    size_t i = 0;
    while (i < route.backends.size()) {
      size_t j = i;
      while (j < route.backends.size() &&
             route.backends[j].priority == route.backends[i].priority)
        ++j;
      std::vector<std::pair<double, const FastBackend*>> keyed;
      for (size_t k = i; k < j; ++k) {
        double w = std::max(route.backends[k].weight, 1e-9);
        keyed.push_back({-std::pow(uni(rng), 1.0 / w), &route.backends[k]});
      }
      std::sort(keyed.begin(), keyed.end(),
                [](const auto& a, const auto& b) { return a.first < b.first; });
      for (auto& kv : keyed) order.push_back(kv.second);
      i = j;
    }
    return order;
  }

  // returns 0 done+keepalive, 1 retriable failure, 2 done+close
  int try_backend(const HttpHead& req, const std::string& body,
                  const aigw_core::Scan& sc, const FastRoute& route,
                  const FastBackend& be, bool stream, int64_t gpu_tokens,
                  int64_t t0, int32_t cache_slot = -1, uint64_t cache_fp = 0) {
    // per-try body: splice from ORIGINAL bytes each attempt
    std::string out_body;
    const std::string* send_body = &body;
    if (!be.model_override.empty() && sc.model_ve > sc.model_vs) {
      out_body = splice_model(body, sc.model_vs, sc.model_ve, be.model_override);
      send_body = &out_body;
    }
    if (stream && route.has_costs) {
      if (send_body == &body) out_body = body, send_body = &out_body;
      splice_include_usage(&out_body);
    }

    std::string head;
    head.reserve(512 + req.head_len);
    head += "POST ";
    if (be.azure) {
      // deployments-API rewrite; the deployment name is the effective
      // model (override wins), suffix derived from the OpenAI path
      const std::string& model_eff =
          be.model_override.empty() ? sc.model : be.model_override;
      const char* suffix = req.path == "/v1/embeddings" ? "embeddings"
                           : req.path == "/v1/completions"
                               ? "completions"
                               : "chat/completions";
      head += "/openai/deployments/" + model_eff + "/" + suffix +
              "?api-version=" +
              (be.azure_api_version.empty() ? "2025-01-01-preview"
                                            : be.azure_api_version);
    } else {
      head += req.path;
    }
    head += " HTTP/1.1\r\nhost: ";
    head += be.host;
    if (be.port != 80 && be.port != 443) head += ":" + std::to_string(be.port);
    head += "\r\ncontent-length: " + std::to_string(send_body->size());
    head += "\r\ncontent-type: application/json\r\n";
    for (const auto& h : req.headers) {
      if (banned_upstream_header(h.name)) continue;
      head += h.name;
      head += ": ";
      head += h.value;
      head += "\r\n";
    }
    if (be.azure && (!be.bearer.empty() || !be.api_key_file.empty())) {
      head += "api-key: ";
      head += srv_->resolve_bearer(be);
      head += "\r\n";
    } else if (!be.bearer.empty() || !be.api_key_file.empty()) {
      head += "authorization: Bearer ";
      head += srv_->resolve_bearer(be);
      head += "\r\n";
    } else {
      // propagate client Authorization for auth-less backends
      const std::string* auth = req.get("authorization");
      if (auth) head += "authorization: " + *auth + "\r\n";
    }
    head += "\r\n";

    int ufd = srv_->pool_->acquire(be.host, be.port, be.timeout_s);
    if (ufd < 0) return 1;
    if (!write_two(ufd, head, *send_body)) {
      ::close(ufd);
      return 1;
    }

    // response head
    std::string rbuf;
    char tmp[65536];
    size_t head_end;
    for (;;) {
      head_end = rbuf.find("\r\n\r\n");
      if (head_end != std::string::npos) break;
      if (rbuf.size() > kMaxHead) {
        ::close(ufd);
        return 1;
      }
      ssize_t r = read_some(ufd, tmp, sizeof(tmp));
      if (r <= 0) {
        ::close(ufd);
        return 1;
      }
      rbuf.append(tmp, (size_t)r);
    }
    HttpHead resp;
    if (!parse_head(rbuf, head_end, false, &resp)) {
      ::close(ufd);
      return 1;
    }
    rbuf.erase(0, resp.head_len);
    bool up_close = false;
    {
      const std::string* c = resp.get("connection");
      if (c && lower(*c).find("close") != std::string::npos) up_close = true;
    }

    static const int kRetriable[] = {429, 500, 502, 503, 504};
    bool retriable = false;
    for (int s : kRetriable) retriable |= (resp.status == s);

    int64_t length = -1;
    Framing fr = response_framing(resp, &length);

    if (resp.status >= 400) {
      std::string err_body;
      if (!read_full_body(ufd, fr, length, rbuf, &err_body)) {
        ::close(ufd);
        return 1;
      }
      srv_->pool_->release(be.host, be.port, ufd, !up_close && fr != Framing::kClose);
      if (retriable) return 1;  // caller keeps the cache slot for the next try
      if (cache_slot >= 0) srv_->gpu_direct_->cache_release_slot(cache_slot);
      srv_->stats_.responses_4xx++;
      record_latency(t0);
      return forward_buffered(resp, err_body) ? 0 : 2;
    }

    if (stream) {
      int rc = relay_stream(resp, fr, length, rbuf, ufd, be, up_close,
                            gpu_tokens, t0);
      return rc;
    }

    std::string resp_body;
    if (!read_full_body(ufd, fr, length, rbuf, &resp_body)) {
      ::close(ufd);
      return 1;
    }
    srv_->pool_->release(be.host, be.port, ufd, !up_close && fr != Framing::kClose);
    Usage u;
    if (!resp.get("content-encoding"))
      extract_usage(resp_body.data(), resp_body.size(), &u);
    finish_usage(u, gpu_tokens);
    if (cache_slot >= 0) {
      if (resp_body.size() < (2u << 20)) {
        // value first, then the index row: a lookup racing the insert
        // can misscore the half-written vector but never map a row to a
        // stale value (same cap as the Python cache)
        std::string v;
        v.reserve(9 + resp_body.size());
        v.append(reinterpret_cast<const char*>(&cache_fp), 8);
        v.push_back('U');
        v += resp_body;
        std::lock_guard<std::mutex> lk(srv_->cache_mu_);
        long long row = srv_->gpu_direct_->cache_insert_slot(cache_slot);
        if (row >= 0 && (size_t)row < srv_->cache_values_.size())
          srv_->cache_values_[(size_t)row] = std::move(v);
      } else {
        srv_->gpu_direct_->cache_release_slot(cache_slot);
      }
    }
    srv_->stats_.responses_2xx++;
    record_latency(t0);
    return forward_buffered(resp, resp_body) ? 0 : 2;
  }

  // read a complete framed body from the upstream, decoded (de-chunked)
  bool read_full_body(int fd, Framing fr, int64_t length, std::string& rbuf,
                      std::string* out) {
    char tmp[65536];
    if (fr == Framing::kLength) {
      out->assign(rbuf, 0, std::min((size_t)length, rbuf.size()));
      rbuf.erase(0, out->size());
      while ((int64_t)out->size() < length) {
        ssize_t r = read_some(fd, tmp, sizeof(tmp));
        if (r <= 0) return false;
        out->append(tmp, (size_t)r);
      }
      if ((int64_t)out->size() > length) {
        rbuf.insert(0, out->substr((size_t)length));
        out->resize((size_t)length);
      }
      return true;
    }
    if (fr == Framing::kChunked) {
      ChunkedParser cp;
      auto tap = [&](const char* p, size_t n) { out->append(p, n); };
      cp.feed(rbuf.data(), rbuf.size(), tap);
      rbuf.clear();
      while (!cp.done() && !cp.error()) {
        ssize_t r = read_some(fd, tmp, sizeof(tmp));
        if (r <= 0) return false;
        cp.feed(tmp, (size_t)r, tap);
      }
      return cp.done();
    }
    // close-delimited
    out->assign(rbuf);
    rbuf.clear();
    for (;;) {
      ssize_t r = read_some(fd, tmp, sizeof(tmp));
      if (r < 0) return false;
      if (r == 0) return true;
      out->append(tmp, (size_t)r);
      if (out->size() > kMaxBody) return false;
    }
  }

  // forward a buffered upstream response (status + filtered headers +
  // explicit content-length + body); returns keep-alive viability
  bool forward_buffered(const HttpHead& resp, const std::string& body) {
    std::string head = "HTTP/1.1 " + std::to_string(resp.status) + " X\r\n";
    for (const auto& h : resp.headers) {
      if (h.name == "connection" || h.name == "keep-alive" ||
          h.name == "transfer-encoding" || h.name == "content-length")
        continue;
      head += h.name;
      head += ": ";
      head += h.value;
      head += "\r\n";
    }
    head += "content-length: " + std::to_string(body.size()) + "\r\n\r\n";
    srv_->stats_.bytes_out += head.size() + body.size();
    return write_two(fd_, head, body);
  }

  // relay a streamed response with its upstream framing preserved and a
  // usage tap on the payload (SSE re-chunk parity: the bytes are already
  // SSE; this path forwards them unmodified, per OpenAI passthrough)
  int relay_stream(const HttpHead& resp, Framing fr, int64_t length,
                   std::string& rbuf, int ufd, const FastBackend& be,
                   bool up_close, int64_t gpu_tokens, int64_t t0) {
    std::string head = "HTTP/1.1 " + std::to_string(resp.status) + " X\r\n";
    for (const auto& h : resp.headers) {
      if (h.name == "connection" || h.name == "keep-alive") continue;
      // content-length dropped on streams (A.8) unless length-framed relay
      if (h.name == "content-length" && fr != Framing::kLength) continue;
      head += h.name;
      head += ": ";
      head += h.value;
      head += "\r\n";
    }
    bool client_close = fr == Framing::kClose;
    if (client_close) head += "connection: close\r\n";
    head += "\r\n";
    if (!write_all(fd_, head)) {
      ::close(ufd);
      return 2;
    }
    srv_->stats_.bytes_out += head.size();

    Usage u;
    bool scan_usage = !resp.get("content-encoding");
    ChunkedParser cp;
    char tmp[65536];
    int64_t remaining = length;
    bool done = false, up_err = false, client_err = false;
    auto tap = [&](const char* p, size_t n) {
      if (scan_usage && memmem(p, n, "usage", 5)) extract_usage(p, n, &u);
    };
    // first: whatever arrived with the head
    auto pump = [&](const char* p, size_t n) -> bool {
      if (n == 0) return true;
      if (fr == Framing::kChunked) {
        cp.feed(p, n, tap);
        if (cp.error()) up_err = true;
        if (cp.done()) done = true;
      } else {
        tap(p, n);
        if (fr == Framing::kLength) {
          remaining -= (int64_t)n;
          if (remaining <= 0) done = true;
        }
      }
      srv_->stats_.bytes_out += n;
      if (!write_all(fd_, p, n)) {
        client_err = true;
        return false;
      }
      return true;
    };
    pump(rbuf.data(), rbuf.size());
    rbuf.clear();
    while (!done && !up_err && !client_err) {
      ssize_t r = read_some(ufd, tmp, sizeof(tmp));
      if (r < 0) {
        up_err = true;
        break;
      }
      if (r == 0) {
        if (fr == Framing::kClose) done = true;
        else up_err = true;
        break;
      }
      if (!pump(tmp, (size_t)r)) break;
    }
    finish_usage(u, gpu_tokens);
    record_latency(t0);
    if (up_err || client_err || fr == Framing::kClose || up_close) {
      ::close(ufd);
    } else {
      srv_->pool_->release(be.host, be.port, ufd, true);
    }
    if (client_err) return 2;
    srv_->stats_.responses_2xx++;
    // a mid-stream upstream failure after bytes went out cannot fall
    // back; the client sees truncated framing (stream cut) — close
    if (up_err) return 2;
    return client_close ? 2 : 0;
  }

  // relay any non-hot request to the loopback Python fallback app
  bool fallback(const HttpHead& req, const std::string& body) {
    srv_->stats_.fallback++;
    if (srv_->fallback_port_ == 0)
      return simple_reply(404, "not_found", "not found", false);
    int ufd = srv_->pool_->acquire(srv_->fallback_host_, srv_->fallback_port_, 300.0);
    if (ufd < 0)
      return simple_reply(502, "upstream_error", "fallback unavailable", false);
    std::string head = req.method + " " + req.path + " HTTP/1.1\r\n";
    const std::string* host = req.get("host");
    head += "host: " + (host ? *host : "127.0.0.1") + "\r\n";
    head += "content-length: " + std::to_string(body.size()) + "\r\n";
    for (const auto& h : req.headers) {
      if (h.name == "host" || h.name == "content-length" ||
          h.name == "connection" || h.name == "transfer-encoding")
        continue;
      head += h.name + ": " + h.value + "\r\n";
    }
    head += "\r\n";
    if (!write_two(ufd, head, body)) {
      ::close(ufd);
      return simple_reply(502, "upstream_error", "fallback write failed", false);
    }
    std::string rbuf;
    char tmp[65536];
    size_t head_end;
    for (;;) {
      head_end = rbuf.find("\r\n\r\n");
      if (head_end != std::string::npos) break;
      ssize_t r = read_some(ufd, tmp, sizeof(tmp));
      if (r <= 0 || rbuf.size() > kMaxHead) {
        ::close(ufd);
        return simple_reply(502, "upstream_error", "fallback read failed", false);
      }
      rbuf.append(tmp, (size_t)r);
    }
    HttpHead resp;
    if (!parse_head(rbuf, head_end, false, &resp)) {
      ::close(ufd);
      return simple_reply(502, "upstream_error", "fallback bad response", false);
    }
    rbuf.erase(0, resp.head_len);
    int64_t length = -1;
    Framing fr = response_framing(resp, &length);
    bool up_close = false;
    {
      const std::string* c = resp.get("connection");
      if (c && lower(*c).find("close") != std::string::npos) up_close = true;
    }
    // relay with framing preserved (chunked streams stay streams)
    std::string chead = "HTTP/1.1 " + std::to_string(resp.status) + " X\r\n";
    for (const auto& h : resp.headers) {
      if (h.name == "connection" || h.name == "keep-alive") continue;
      chead += h.name + ": " + h.value + "\r\n";
    }
    bool client_close = fr == Framing::kClose;
    if (client_close) chead += "connection: close\r\n";
    chead += "\r\n";
    if (!write_all(fd_, chead)) {
      ::close(ufd);
      return false;
    }
    srv_->stats_.bytes_out += chead.size();
    ChunkedParser cp;
    int64_t remaining = length;
    bool done = (fr == Framing::kLength && length == 0), up_err = false;
    auto pump = [&](const char* p, size_t n) -> bool {
      if (n == 0) return true;
      if (fr == Framing::kChunked) {
        cp.feed(p, n, [](const char*, size_t) {});
        if (cp.error()) up_err = true;
        if (cp.done()) done = true;
      } else if (fr == Framing::kLength) {
        remaining -= (int64_t)n;
        if (remaining <= 0) done = true;
      }
      srv_->stats_.bytes_out += n;
      return write_all(fd_, p, n);
    };
    if (!pump(rbuf.data(), rbuf.size())) {
      ::close(ufd);
      return false;
    }
    rbuf.clear();
    while (!done && !up_err) {
      ssize_t r = read_some(ufd, tmp, sizeof(tmp));
      if (r < 0) {
        up_err = true;
        break;
      }
      if (r == 0) {
        if (fr == Framing::kClose) done = true;
        else up_err = true;
        break;
      }
      if (!pump(tmp, (size_t)r)) {
        ::close(ufd);
        return false;
      }
    }
    if (resp.status < 400) srv_->stats_.responses_2xx++;
    else if (resp.status < 500) srv_->stats_.responses_4xx++;
    else srv_->stats_.responses_5xx++;
    if (up_err || fr == Framing::kClose || up_close)
      ::close(ufd);
    else
      srv_->pool_->release(srv_->fallback_host_, srv_->fallback_port_, ufd, true);
    return !client_close && !up_err;
  }

  // collect the deferred admission count (enqueued pre-dispatch): by
  // now the batch has usually completed while the upstream round-trip
  // was in flight, so this wait is ~zero
  int64_t resolve_pending_count() {
    if (pending_count_ == nullptr) return 0;
    auto cl = srv_->gpu_direct_->collect(pending_count_);
    pending_count_.reset();
    srv_->gpu_direct_->cache_release_slot(cl.slot);
    srv_->stats_.gpu_tokens += (uint64_t)cl.tokens;
    return cl.tokens;
  }

  void finish_usage(Usage& u, int64_t gpu_tokens) {
    if (pending_count_ != nullptr) {
      int64_t t = resolve_pending_count();
      if (gpu_tokens == 0) gpu_tokens = t;
    }
    if (u.total == 0 && gpu_tokens > 0) {
      // upstream reported no usage: gateway's own GPU-tokenized count
      u.input = gpu_tokens;
      u.total = gpu_tokens + u.output;
    }
    srv_->stats_.input_tokens += (uint64_t)u.input;
    srv_->stats_.output_tokens += (uint64_t)u.output;
    srv_->stats_.total_tokens += (uint64_t)u.total;
    srv_->rl_charge(u);
  }

  void record_latency(int64_t t0) {
    int64_t us = now_us() - t0;
    if (us < 1) us = 1;
    int b = 63 - __builtin_clzll((uint64_t)us);
    if (b > 31) b = 31;
    srv_->stats_.latency_us_log2[b]++;
  }

  bool raw_reply(int status, const char* ctype, const std::string& body) {
    std::string head = "HTTP/1.1 " + std::to_string(status) + " X\r\ncontent-type: " +
                       ctype + "\r\ncontent-length: " + std::to_string(body.size()) +
                       "\r\n\r\n";
    srv_->stats_.bytes_out += head.size() + body.size();
    return write_two(fd_, head, body);
  }

  bool simple_reply(int status, const char* type, const std::string& msg,
                    bool close_after, const std::string& extra_headers = "") {
    std::string body = "{\"type\":\"error\",\"error\":{\"type\":\"";
    body += type;
    body += "\",\"code\":\"";
    body += std::to_string(status);
    body += "\",\"message\":\"";
    for (char c : msg) {
      if (c == '"' || c == '\\') body.push_back('\\');
      if ((unsigned char)c >= 0x20) body.push_back(c);
    }
    body += "\"}}";
    std::string head = "HTTP/1.1 " + std::to_string(status) +
                       " X\r\ncontent-type: application/json\r\ncontent-length: " +
                       std::to_string(body.size()) + "\r\n" + extra_headers;
    if (close_after) head += "connection: close\r\n";
    head += "\r\n";
    srv_->stats_.bytes_out += head.size() + body.size();
    bool ok = write_two(fd_, head, body);
    return ok && !close_after;
  }

  FastServer* srv_;
  int fd_;
  // deferred admission count for the CURRENT request (enqueued before
  // upstream dispatch, collected in finish_usage)
  std::shared_ptr<DirectGpuBatcher::Waiter2> pending_count_;

  friend class FastServer;
};


// -------- native bench harness ----------------------------------------------
//
// The serving benchmark needs a load generator and a mock upstream whose
// capacity comfortably exceeds the gateway's; Python asyncio clients cap
// near ~2k req/s per process for 25 KiB bodies (measured: fast-front
// throughput scaled linearly with loadgen count while the server sat
// idle), so the harness itself is native: thread-per-connection closed
// loops, same machinery as the server.

int FastMock::start(const std::string& host, const std::string& response) {
    response_ = response;
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("mock socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = 0;
    inet_pton(AF_INET, host.c_str(), &addr.sin_addr);
    if (::bind(listen_fd_, (struct sockaddr*)&addr, sizeof(addr)) != 0 ||
        ::listen(listen_fd_, 4096) != 0)
      throw std::runtime_error("mock bind/listen failed");
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, (struct sockaddr*)&addr, &alen);
    acceptor_ = std::thread([this] {
      for (;;) {
        int fd = ::accept(listen_fd_, nullptr, nullptr);
        if (fd < 0) return;
        set_nodelay(fd);
        set_timeout(fd, 300.0);
        std::thread([this, fd] { serve(fd); }).detach();
      }
    });
    return ntohs(addr.sin_port);
}

void FastMock::stop() {
    if (listen_fd_ >= 0) {
      ::shutdown(listen_fd_, SHUT_RDWR);
      ::close(listen_fd_);
      listen_fd_ = -1;
    }
    if (acceptor_.joinable()) acceptor_.join();
}

uint64_t FastMock::requests() const { return served_.load(); }

void FastMock::serve(int fd) {
    std::string buf;
    char tmp[65536];
    for (;;) {
      size_t head_end;
      for (;;) {
        head_end = buf.find("\r\n\r\n");
        if (head_end != std::string::npos) break;
        ssize_t r = read_some(fd, tmp, sizeof(tmp));
        if (r <= 0) { ::close(fd); return; }
        buf.append(tmp, (size_t)r);
      }
      int64_t clen = 0;
      {
        size_t p = buf.find("content-length:");
        if (p == std::string::npos) p = buf.find("Content-Length:");
        if (p != std::string::npos && p < head_end)
          clen = atoll(buf.c_str() + p + 15);
      }
      while (buf.size() < head_end + 4 + (size_t)clen) {
        ssize_t r = read_some(fd, tmp, sizeof(tmp));
        if (r <= 0) { ::close(fd); return; }
        buf.append(tmp, (size_t)r);
      }
      buf.erase(0, head_end + 4 + (size_t)clen);
      if (!write_all(fd, response_)) { ::close(fd); return; }
      served_++;
    }
}

// closed-loop load: `connections` threads, each one keep-alive connection
// issuing `per_conn` sequential POSTs to host:port/path, cycling the
// body pool (one body = fixed payload; many = semantic-cache hit mix).
LoadResult run_load_pool(const std::string& host, uint16_t port,
                         const std::string& path,
                         const std::vector<std::string>& bodies,
                         int connections, int per_conn) {
  std::vector<std::string> reqs;
  reqs.reserve(bodies.size());
  for (const auto& body : bodies)
    reqs.push_back("POST " + path + " HTTP/1.1\r\nhost: bench\r\n"
                   "content-type: application/json\r\ncontent-length: " +
                   std::to_string(body.size()) + "\r\n\r\n" + body);
  std::vector<std::thread> threads;
  std::vector<std::vector<int64_t>> lat(connections);
  std::atomic<uint64_t> errors{0};
  std::atomic<uint64_t> completed{0};
  int64_t t0 = now_us();
  for (int c = 0; c < connections; ++c) {
    threads.emplace_back([&, c] {
      lat[c].reserve(per_conn);
      int fd = -1;
      std::string buf;
      char tmp[65536];
      for (int i = 0; i < per_conn; ++i) {
        const std::string& req = reqs[((size_t)i * connections + c) % reqs.size()];
        int64_t r0 = now_us();
        if (fd < 0) {
          fd = tcp_connect(host, port, 120.0);
          if (fd < 0) { errors++; continue; }
          buf.clear();
        }
        if (!write_all(fd, req)) {
          ::close(fd); fd = -1; errors++; continue;
        }
        // read one response (content-length framing; the gateway and the
        // mock both answer length-framed unary bodies)
        size_t head_end;
        bool fail = false;
        for (;;) {
          head_end = buf.find("\r\n\r\n");
          if (head_end != std::string::npos) break;
          ssize_t r = read_some(fd, tmp, sizeof(tmp));
          if (r <= 0) { fail = true; break; }
          buf.append(tmp, (size_t)r);
        }
        if (fail) { ::close(fd); fd = -1; errors++; continue; }
        bool chunked = false;
        int64_t clen = 0;
        {
          size_t p = buf.find("content-length:");
          if (p == std::string::npos) p = buf.find("Content-Length:");
          if (p != std::string::npos && p < head_end)
            clen = atoll(buf.c_str() + p + 15);
          size_t t = buf.find("transfer-encoding: chunked");
          if (t == std::string::npos)
            t = buf.find("Transfer-Encoding: chunked");
          chunked = t != std::string::npos && t < head_end;
        }
        bool ok = buf.compare(9, 3, "200") == 0;
        if (chunked) {
          // streamed response (SSE): consume chunked framing to the
          // terminal chunk so the connection stays reusable
          buf.erase(0, head_end + 4);
          ChunkedParser cp;
          cp.feed(buf.data(), buf.size(), [](const char*, size_t) {});
          buf.clear();
          while (!cp.done() && !cp.error()) {
            ssize_t r = read_some(fd, tmp, sizeof(tmp));
            if (r <= 0) { fail = true; break; }
            cp.feed(tmp, (size_t)r, [](const char*, size_t) {});
          }
          if (fail || cp.error()) { ::close(fd); fd = -1; errors++; continue; }
        } else {
          while (buf.size() < head_end + 4 + (size_t)clen) {
            ssize_t r = read_some(fd, tmp, sizeof(tmp));
            if (r <= 0) { fail = true; break; }
            buf.append(tmp, (size_t)r);
          }
          if (fail) { ::close(fd); fd = -1; errors++; continue; }
          buf.erase(0, head_end + 4 + (size_t)clen);
        }
        if (!ok) { errors++; continue; }
        completed++;
        lat[c].push_back(now_us() - r0);
      }
      if (fd >= 0) ::close(fd);
    });
  }
  for (auto& t : threads) t.join();
  LoadResult out;
  out.elapsed_s = (double)(now_us() - t0) / 1e6;
  out.completed = completed.load();
  out.errors = errors.load();
  std::vector<int64_t> all;
  for (auto& v : lat) all.insert(all.end(), v.begin(), v.end());
  if (!all.empty()) {
    std::sort(all.begin(), all.end());
    out.p50_ms = (double)all[all.size() / 2] / 1e3;
    out.p99_ms = (double)all[std::min(all.size() - 1,
                                      (size_t)((double)all.size() * 0.99))] / 1e3;
  }
  return out;
}

LoadResult run_load(const std::string& host, uint16_t port,
                    const std::string& path, const std::string& body,
                    int connections, int per_conn) {
  return run_load_pool(host, port, path, {body}, connections, per_conn);
}

// -------- FastServer --------------------------------------------------------

FastServer::FastServer() : pool_(new UpstreamPool) {}

FastServer::~FastServer() { stop(); }

static void normalize_route(FastRoute& r) {
  std::stable_sort(r.backends.begin(), r.backends.end(),
                   [](const FastBackend& a, const FastBackend& b) {
                     return a.priority < b.priority;
                   });
  if (r.model_match.empty()) r.catch_all = true;
}

void FastServer::add_route(FastRoute r) {
  normalize_route(r);
  auto next = std::make_shared<std::vector<FastRoute>>(*routes_);
  next->push_back(std::move(r));
  routes_ = std::move(next);
}

void FastServer::swap_routes(std::vector<FastRoute> routes) {
  for (auto& r : routes) normalize_route(r);
  // atomic shared_ptr store: handlers snapshot the table per request
  std::atomic_store(&routes_,
                    std::shared_ptr<const std::vector<FastRoute>>(
                        std::make_shared<const std::vector<FastRoute>>(
                            std::move(routes))));
}

void FastServer::add_rate_rule(const RateRule& r) {
  auto rs = std::make_unique<RuleState>();
  rs->rule = r;
  rs->window_start_ms = now_ms();
  rules_.push_back(std::move(rs));
}

void FastServer::set_fallback(const std::string& host, uint16_t port) {
  fallback_host_ = host;
  fallback_port_ = port;
}

void FastServer::enable_gpu(const std::string& socket_path, int window_us,
                            int max_batch) {
  auto client = std::make_unique<GpuAdmissionClient>();
  if (!client->start(socket_path, window_us, max_batch)) {
    throw std::runtime_error("cannot connect GPU admission service at " + socket_path);
  }
  gpu_.push_back(std::move(client));
}

std::vector<uint64_t> FastServer::gpu_direct_stats() const {
  if (gpu_direct_ == nullptr) return {0, 0, 0, 0, 0, 0, 0, 0, 0};
  return {gpu_direct_->stats_batches.load(), gpu_direct_->stats_texts.load(),
          gpu_direct_->stats_time_us.load(), gpu_direct_->stats_max_us.load(),
          gpu_direct_->stats_errors.load(),
          gpu_direct_->stats_pack_us.load(),
          gpu_direct_->stats_submit_us.load(),
          gpu_direct_->stats_wait_us.load(),
          gpu_direct_->stats_fulfill_us.load()};
}

bool FastServer::gpu_enabled() const {
  return gpu_direct_ != nullptr || !gpu_.empty();
}

int64_t FastServer::gpu_count(const std::string& text) {
  // in-process HIP admission wins when available; the UDS hosts remain
  // for deployments where the gateway process must not own the device
  if (gpu_direct_ != nullptr) return gpu_direct_->count_text(text);
  size_t i = (size_t)(gpu_rr_.fetch_add(1, std::memory_order_relaxed) % gpu_.size());
  return gpu_[i]->count_text(text);
}

bool FastServer::enable_gpu_direct_cache(const uint16_t* emb, int vocab,
                                         const uint16_t* proj, int dim,
                                         long long capacity, float threshold,
                                         bool fp8) {
  if (gpu_direct_ == nullptr) return false;
  if (!gpu_direct_->init_cache(emb, vocab, proj, dim, capacity, threshold, fp8))
    return false;
  cache_values_.assign((size_t)capacity, std::string());
  return true;
}

void FastServer::enable_gpu_direct(const long long* htab_keys,
                                   const int32_t* htab_rank, int htab_n,
                                   int max_batch, size_t max_batch_bytes,
                                   int max_req, int device) {
  auto b = std::make_unique<DirectGpuBatcher>();
  if (!b->start(htab_keys, htab_rank, htab_n, max_batch, max_batch_bytes,
                max_req, device))
    throw std::runtime_error(
        "direct GPU admission init failed (no device visible?)");
  gpu_direct_ = std::move(b);
}

int FastServer::start(const std::string& host, uint16_t port) {
  listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
  if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
  int one = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_port = htons(port);
  if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
    addr.sin_addr.s_addr = INADDR_ANY;
  if (::bind(listen_fd_, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
    ::close(listen_fd_);
    listen_fd_ = -1;
    throw std::runtime_error("bind failed: " + std::string(strerror(errno)));
  }
  if (::listen(listen_fd_, 4096) != 0) {
    ::close(listen_fd_);
    listen_fd_ = -1;
    throw std::runtime_error("listen failed");
  }
  socklen_t alen = sizeof(addr);
  getsockname(listen_fd_, (struct sockaddr*)&addr, &alen);
  acceptor_ = std::thread([this] { accept_loop(); });
  return ntohs(addr.sin_port);
}

void FastServer::accept_loop() {
  for (;;) {
    int fd = ::accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) {
      if (errno == EINTR) continue;
      return;  // listener closed (stop)
    }
    if (stopping_) {
      ::close(fd);
      return;
    }
    set_nodelay(fd);
    set_timeout(fd, 300.0);
    {
      std::lock_guard<std::mutex> lk(conn_mu_);
      if (conn_fds_.size() >= 8192) {
        ::close(fd);
        continue;
      }
      conn_fds_.insert(fd);
    }
    stats_.active_connections++;
    std::thread([this, fd] {
      ConnHandler(this, fd).run();
      {
        std::lock_guard<std::mutex> lk(conn_mu_);
        conn_fds_.erase(fd);
      }
      ::close(fd);
      stats_.active_connections--;
    }).detach();
  }
}

int FastServer::drain(double drain_s) {
  // fail new connections first (the LB health check sees the listener
  // gone), let in-flight requests finish
  if (listen_fd_ >= 0) {
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    listen_fd_ = -1;
  }
  if (acceptor_.joinable()) acceptor_.join();
  int waited_ms = 0;
  int limit_ms = (int)(drain_s * 1000.0);
  while (stats_.active_connections.load() > 0 && waited_ms < limit_ms) {
    std::this_thread::sleep_for(std::chrono::milliseconds(10));
    waited_ms += 10;
  }
  return (int)stats_.active_connections.load();
}

void FastServer::stop() {
  if (stopping_.exchange(true)) return;
  if (listen_fd_ >= 0) {
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    listen_fd_ = -1;
  }
  if (acceptor_.joinable()) acceptor_.join();
  // nudge live connections to finish: shut down their sockets; the
  // per-connection threads exit on the read error
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    for (int fd : conn_fds_) ::shutdown(fd, SHUT_RDWR);
  }
  for (int i = 0; i < 200 && stats_.active_connections.load() > 0; ++i)
    std::this_thread::sleep_for(std::chrono::milliseconds(10));
  if (gpu_direct_) gpu_direct_->stop();
  for (auto& g : gpu_) g->stop();
  pool_->close_all();
}

std::string FastServer::resolve_bearer(const FastBackend& be) {
  if (be.api_key_file.empty()) return be.bearer;
  // mtime-cached file credential (rotation without restart)
  std::lock_guard<std::mutex> lk(cred_mu_);
  auto& fc = file_creds_[be.api_key_file];
  int64_t now = now_ms();
  if (now - fc.checked_ms > 1000 || fc.value.empty()) {
    fc.checked_ms = now;
    FILE* f = fopen(be.api_key_file.c_str(), "rb");
    if (f) {
      char buf[4096];
      size_t n = fread(buf, 1, sizeof(buf) - 1, f);
      fclose(f);
      while (n > 0 && (buf[n - 1] == '\n' || buf[n - 1] == '\r' ||
                       buf[n - 1] == ' '))
        --n;
      fc.value.assign(buf, n);
    }
  }
  return fc.value.empty() ? be.bearer : fc.value;
}

// -------- rate limiting (fixed-window, cross-shard syncable) ----------------

void FastServer::rl_roll(RuleState& rs, int64_t now) {
  int64_t ws = rs.window_start_ms.load(std::memory_order_relaxed);
  if (now - ws >= (int64_t)(rs.rule.window_s * 1000.0)) {
    if (rs.window_start_ms.compare_exchange_strong(ws, now)) {
      rs.local_spent = 0;
      rs.remote_spent = 0;
    }
  }
}

bool FastServer::rl_check(double* retry_after_s, std::string* rule_name) {
  int64_t now = now_ms();
  for (auto& rsp : rules_) {
    auto& rs = *rsp;
    rl_roll(rs, now);
    int64_t total = rs.local_spent.load(std::memory_order_relaxed) +
                    rs.remote_spent.load(std::memory_order_relaxed);
    if (total >= rs.rule.limit) {
      double elapsed = (double)(now - rs.window_start_ms.load()) / 1000.0;
      *retry_after_s = std::max(rs.rule.window_s - elapsed, 0.0);
      *rule_name = rs.rule.name;
      return false;
    }
  }
  return true;
}

void FastServer::rl_charge(const Usage& u) {
  if (rules_.empty()) return;
  int64_t now = now_ms();
  for (auto& rsp : rules_) {
    auto& rs = *rsp;
    int64_t cost = rs.rule.metadata_key == 1   ? u.input
                   : rs.rule.metadata_key == 2 ? u.output
                                               : u.total;
    if (!cost) continue;
    rl_roll(rs, now);
    rs.local_spent += cost;
    rs.pending_delta += cost;
  }
}

std::vector<int64_t> FastServer::rl_collect_deltas() {
  std::vector<int64_t> out;
  out.reserve(rules_.size());
  for (auto& rsp : rules_) out.push_back(rsp->pending_delta.exchange(0));
  return out;
}

void FastServer::rl_apply_remote(const std::vector<int64_t>& others_spend) {
  for (size_t i = 0; i < rules_.size() && i < others_spend.size(); ++i)
    rules_[i]->remote_spent += others_spend[i];
}

std::vector<int64_t> FastServer::rl_local_spent() const {
  std::vector<int64_t> out;
  for (const auto& rsp : rules_) out.push_back(rsp->local_spent.load());
  return out;
}

}  // namespace aigw_fast
