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
// GPU admission stubs (admission.hip is hipcc-only; the sanitizer lane
// runs the GPU-off code path)
class GpuAdmissionDirect {};
GpuAdmissionDirect* admission_create(const long long*, const int32_t*, int,
                                     size_t, int, int) {
  return nullptr;
}
bool admission_count(GpuAdmissionDirect*, const char*, size_t, const int64_t*,
                     int, int32_t*) {
  return false;
}
bool admission_init_cache(GpuAdmissionDirect*, const uint16_t*, int,
                          const uint16_t*, int, long long, float, int, bool) {
  return false;
}
bool admission_count_lookup(GpuAdmissionDirect*, const char*, size_t,
                            const int64_t*, int, int32_t*, const int32_t*,
                            int32_t*, float*) {
  return false;
}
long long admission_cache_insert(GpuAdmissionDirect*, int) { return -1; }
bool admission_submit(GpuAdmissionDirect*, int, const char*, size_t,
                      const int64_t*, int, const int32_t*) {
  return false;
}
bool admission_wait(GpuAdmissionDirect*, int, int, int32_t*, int32_t*, float*) {
  return false;
}
void admission_destroy(GpuAdmissionDirect*) {}
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
