// ASan/UBSan + thread stress of the native data-plane server
// (csrc/fastpath.cpp) — the `go test -race` lane analogue for the hot
// path the reference runs through Envoy C++. Built CPU-only by
// scripts/sanitize_native.sh with the GPU admission entry points stubbed
// (GPU-off deployments take exactly this path), exercising: concurrent
// keep-alive clients, hot route swaps mid-traffic, retries to a dead
// backend, streamed chunked relay via the fallback mock, rate-limit
// denials, desync rejections, and drain/stop teardown.

#include <cassert>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "fastpath.h"

namespace aigw_fast {
// Functional FAKE admission (admission.hip is hipcc-only): a 2-set
// in-memory emulation so the sanitizer lane exercises the REAL
// DirectGpuBatcher threading — queue handoff, two-set pipelining,
// waiter fulfillment — with deterministic counts (bytes/3) and a
// sleep standing in for the kernel time.
class GpuAdmissionDirect {
 public:
  struct FakeSet {
    std::vector<int32_t> counts;
    std::vector<char> staging;
    int n_req = 0;
  };
  FakeSet sets[2];
};
GpuAdmissionDirect* admission_create(const long long*, const int32_t*, int,
                                     size_t max_bytes, int, int) {
  auto* a = new GpuAdmissionDirect();
  a->sets[0].staging.resize(max_bytes);
  a->sets[1].staging.resize(max_bytes);
  return a;
}
char* admission_staging(GpuAdmissionDirect* a, int set) {
  return a->sets[set].staging.data();
}
bool admission_submit(GpuAdmissionDirect* a, int set, const char*, size_t n,
                      const int64_t* offsets, int n_req, const int32_t*) {
  auto& s = a->sets[set];
  s.counts.assign((size_t)n_req, 0);
  s.n_req = n_req;
  for (int i = 0; i < n_req; ++i) {
    long long b = offsets[i];
    long long e = (i + 1 < n_req) ? offsets[i + 1] : (long long)n;
    s.counts[(size_t)i] = (int32_t)((e - b) / 3);
  }
  return true;
}
bool admission_wait(GpuAdmissionDirect* a, int set, int n_req,
                    int32_t* counts_out, int32_t* rows_out,
                    float* scores_out) {
  std::this_thread::sleep_for(std::chrono::microseconds(200));
  auto& s = a->sets[set];
  if (s.n_req != n_req) return false;
  for (int i = 0; i < n_req; ++i) {
    counts_out[i] = s.counts[(size_t)i];
    if (rows_out != nullptr) rows_out[i] = -1;
    if (scores_out != nullptr) scores_out[i] = 0.f;
  }
  return true;
}
bool admission_count(GpuAdmissionDirect* a, const char* bytes, size_t n,
                     const int64_t* offsets, int n_req, int32_t* counts_out) {
  return admission_submit(a, 0, bytes, n, offsets, n_req, nullptr) &&
         admission_wait(a, 0, n_req, counts_out, nullptr, nullptr);
}
bool admission_init_cache(GpuAdmissionDirect*, const uint16_t*, int,
                          const uint16_t*, int, long long, float, int, bool) {
  return false;
}
bool admission_count_lookup(GpuAdmissionDirect* a, const char* bytes, size_t n,
                            const int64_t* offsets, int n_req,
                            int32_t* counts_out, const int32_t*,
                            int32_t* rows_out, float* scores_out) {
  return admission_submit(a, 0, bytes, n, offsets, n_req, nullptr) &&
         admission_wait(a, 0, n_req, counts_out, rows_out, scores_out);
}
long long admission_cache_insert(GpuAdmissionDirect*, int) { return -1; }
void admission_destroy(GpuAdmissionDirect* a) { delete a; }
}  // namespace aigw_fast

using namespace aigw_fast;

static FastRoute make_route(const char* name, const char* model, int port,
                            int dead_port = 0) {
  FastRoute r;
  r.name = name;
  r.model_match = model;
  r.retries = 2;
  r.has_costs = true;
  if (dead_port) {
    FastBackend dead;
    dead.name = "dead";
    dead.host = "127.0.0.1";
    dead.port = (uint16_t)dead_port;
    dead.priority = 0;
    r.backends.push_back(dead);
  }
  FastBackend be;
  be.name = "live";
  be.host = "127.0.0.1";
  be.port = (uint16_t)port;
  be.bearer = "sk-test";
  be.priority = 1;
  r.backends.push_back(be);
  return r;
}

int main() {
  std::string body =
      "{\"id\":\"x\",\"object\":\"chat.completion\",\"choices\":[],"
      "\"usage\":{\"prompt_tokens\":5,\"completion_tokens\":2,"
      "\"total_tokens\":7}}";
  std::string canned = "HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
                       "content-length: " +
                       std::to_string(body.size()) + "\r\n\r\n" + body;
  FastMock mock;
  int up_port = mock.start("127.0.0.1", canned);

  FastServer srv;
  srv.add_route(make_route("r1", "model-a", up_port, 1));
  RateRule rule;
  rule.name = "budget";
  rule.limit = 1000000;
  rule.window_s = 3600;
  srv.add_rate_rule(rule);
  int port = srv.start("127.0.0.1", 0);

  std::string payload =
      "{\"model\":\"model-a\",\"messages\":[{\"role\":\"user\","
      "\"content\":\"hello sanitizer world\"}]}";

  // concurrent load while another thread hot-swaps the route table
  std::thread swapper([&] {
    for (int i = 0; i < 50; ++i) {
      std::vector<FastRoute> routes;
      routes.push_back(make_route("r1", "model-a", up_port, i % 2 ? 1 : 0));
      srv.swap_routes(std::move(routes));
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
  });
  LoadResult r = run_load("127.0.0.1", port, "/v1/chat/completions", payload,
                          8, 100);
  swapper.join();
  assert(r.errors == 0);
  assert(r.completed == 800);

  // desync rejections + malformed + unknown model, raw
  {
    LoadResult bad = run_load("127.0.0.1", port, "/v1/chat/completions",
                              "{broken", 2, 5);
    assert(bad.completed == 0 && bad.errors == 10);
  }

  // rate-limit exhaustion: tiny budget, second request denied
  {
    FastServer srv2;
    srv2.add_route(make_route("r", "model-a", up_port));
    RateRule small;
    small.name = "tiny";
    small.limit = 5;
    small.window_s = 3600;
    srv2.add_rate_rule(small);
    int p2 = srv2.start("127.0.0.1", 0);
    LoadResult first = run_load("127.0.0.1", p2, "/v1/chat/completions",
                                payload, 1, 1);
    assert(first.completed == 1);
    LoadResult denied = run_load("127.0.0.1", p2, "/v1/chat/completions",
                                 payload, 1, 1);
    assert(denied.completed == 0 && denied.errors == 1);
    assert(srv2.stats().local_429.load() == 1);
    srv2.stop();
  }

  // pipelined GPU admission batcher under concurrent load: the fake
  // admission routes every request through the real two-set
  // DirectGpuBatcher threading (queue -> pack -> submit/wait rotation
  // -> waiter fulfillment), racing stop() against in-flight batches
  {
    FastServer srv3;
    srv3.add_route(make_route("r", "model-a", up_port));
    std::vector<long long> hk(16, -1);
    std::vector<int32_t> hr(16, -1);
    srv3.enable_gpu_direct(hk.data(), hr.data(), 16, 256, 1 << 20, 512, 0);
    int p3 = srv3.start("127.0.0.1", 0);
    std::thread swapper3([&] {
      for (int i = 0; i < 30; ++i) {
        std::vector<FastRoute> routes;
        routes.push_back(make_route("r", "model-a", up_port));
        srv3.swap_routes(std::move(routes));
        std::this_thread::sleep_for(std::chrono::milliseconds(1));
      }
    });
    LoadResult g = run_load("127.0.0.1", p3, "/v1/chat/completions", payload,
                            8, 100);
    swapper3.join();
    assert(g.errors == 0 && g.completed == 800);
    assert(srv3.stats().gpu_tokens.load() > 0);
    assert(srv3.drain(2.0) == 0);
    srv3.stop();  // joins the batcher with sets possibly in flight
  }

  int left = srv.drain(2.0);
  assert(left == 0);
  srv.stop();
  mock.stop();

  const ServerStats& st = srv.stats();
  std::printf("fastpath sanitize OK: %llu requests, %llu retries, "
              "2xx=%llu 4xx=%llu\n",
              (unsigned long long)st.requests.load(),
              (unsigned long long)st.retries.load(),
              (unsigned long long)st.responses_2xx.load(),
              (unsigned long long)st.responses_4xx.load());
  return 0;
}
