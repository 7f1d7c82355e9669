// aigw fast path — native data-plane server.
//
// The reference splits its data plane between the Envoy C++ proxy (HTTP
// codecs, routing, retries) and a Go extproc (cmd/aigw/run.go:224-231,
// extensionserver/post_translate_modify.go:768); this gateway's round-1
// equivalent ran the whole pipeline in CPython workers, which profiling
// capped at ~117k req/s per node without GPU work. This module moves the
// passthrough hot loop — frame → scan → route → rate-limit → auth →
// upstream dispatch → usage extraction → relay — into native threads
// behind the same RuntimeConfig semantics, with CPython only composing
// configuration and serving cold paths via a loopback fallback.
//
// Concurrency model: one OS thread per client connection (the reference
// runs one goroutine per stream, extproc/server.go:128; at gateway-scale
// connection counts — hundreds per shard, not tens of thousands — the
// thread stack cost is irrelevant and blocking I/O removes an entire
// state-machine layer). An acceptor thread owns the listener; a batcher
// thread coalesces GPU admission calls ACROSS all connections into one
// micro-batch RPC stream to the per-shard GPU service
// (aigw/gpu/service.py), which is strictly better coalescing than the
// round-1 per-worker windows.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <vector>

namespace aigw_fast {

struct FastBackend {
  std::string name;
  std::string host;
  uint16_t port = 0;
  std::string bearer;          // static API-key credential; empty = none
  std::string api_key_file;    // rotated-credential file (mtime-cached)
  std::string model_override;  // modelNameOverride, spliced into the body
  // AzureOpenAI deployments-API rewrite (openai_azureopenai.go): the
  // request path becomes /openai/deployments/<model>/<suffix>?api-version=
  // and the credential header is api-key instead of a bearer
  bool azure = false;
  std::string azure_api_version;
  double weight = 1.0;
  int priority = 0;
  double timeout_s = 60.0;
};

struct FastRoute {
  std::string name;
  std::string model_match;  // exact x-ai-eg-model value; "" = catch-all
  bool catch_all = false;
  int retries = 1;
  bool has_costs = false;  // force include_usage on streams, charge tokens
  bool eligible = true;    // false => relay matched requests to fallback
  std::vector<FastBackend> backends;  // sorted by (priority)
};

struct RateRule {
  std::string name;
  int64_t limit = 0;
  double window_s = 60.0;
  int metadata_key = 0;  // 0=total, 1=input, 2=output tokens
};

struct Usage {
  int64_t input = 0, output = 0, total = 0;
};

struct ServerStats {
  std::atomic<uint64_t> requests{0};
  std::atomic<uint64_t> responses_2xx{0};
  std::atomic<uint64_t> responses_4xx{0};
  std::atomic<uint64_t> responses_5xx{0};
  std::atomic<uint64_t> local_429{0};
  std::atomic<uint64_t> fallback{0};
  std::atomic<uint64_t> retries{0};
  std::atomic<uint64_t> gpu_tokens{0};
  std::atomic<uint64_t> cache_hits{0};
  std::atomic<uint64_t> cache_misses{0};
  std::atomic<uint64_t> input_tokens{0};
  std::atomic<uint64_t> output_tokens{0};
  std::atomic<uint64_t> total_tokens{0};
  std::atomic<uint64_t> bytes_in{0};
  std::atomic<uint64_t> bytes_out{0};
  std::atomic<uint64_t> active_connections{0};
  // log2-microsecond latency histogram (bucket i: [2^i, 2^(i+1)) us)
  std::atomic<uint64_t> latency_us_log2[32] = {};
};

class GpuAdmissionClient;
class GpuAdmissionDirect;  // admission.hip — in-process HIP BPE counting
class DirectGpuBatcher;
class UpstreamPool;

class FastServer {
 public:
  FastServer();
  ~FastServer();

  // configuration (add_route before start; swap_routes any time —
  // in-flight requests keep the table they resolved, the Python
  // config watcher's swap semantics)
  void add_route(FastRoute r);
  void swap_routes(std::vector<FastRoute> routes);
  void add_rate_rule(const RateRule& r);
  void set_fallback(const std::string& host, uint16_t port);
  // may be called multiple times: each socket is one admission-host
  // process (one GPU context each); batches round-robin across them
  void enable_gpu(const std::string& socket_path, int window_us, int max_batch);
  // in-process HIP admission (preferred): the gateway owns one stream
  // and launches the BPE kernels itself — no IPC, no extra GPU context.
  // Throws when no GPU is visible or init fails.
  void enable_gpu_direct(const long long* htab_keys, const int32_t* htab_rank,
                         int htab_n, int max_batch, size_t max_batch_bytes,
                         int max_req, int device);
  // native semantic cache (requires enable_gpu_direct first): embedding
  // table + projection weights, bf16 index ring of `capacity` rows.
  bool enable_gpu_direct_cache(const uint16_t* emb, int vocab,
                               const uint16_t* proj, int dim,
                               long long capacity, float threshold,
                               bool fp8);

  // lifecycle
  int start(const std::string& host, uint16_t port);  // returns bound port
  // graceful drain: stop accepting, wait up to drain_s for in-flight
  // connections to finish, then force-close; returns connections still
  // open at the deadline (0 = clean)
  int drain(double drain_s);
  void stop();

  const ServerStats& stats() const { return stats_; }
  // direct-admission batcher counters: {batches, texts, time_us, max_us,
  // errors, pack_us, submit_us, wait_us, fulfill_us} — the last four are
  // the batcher thread's per-phase wall-time sums (pipeline diagnosis);
  // zeros when direct admission is off
  std::vector<uint64_t> gpu_direct_stats() const;
  // cross-shard rate-limit sync hooks (aigw.parallel.StateSync bridge)
  std::vector<int64_t> rl_collect_deltas();
  void rl_apply_remote(const std::vector<int64_t>& others_spend);
  std::vector<int64_t> rl_local_spent() const;

  std::string resolve_bearer(const FastBackend& be);

 private:
  friend class ConnHandler;
  struct FileCredState {
    int64_t checked_ms = 0;
    std::string value;
  };
  struct RuleState {
    RateRule rule;
    std::atomic<int64_t> window_start_ms{0};
    std::atomic<int64_t> local_spent{0};
    std::atomic<int64_t> remote_spent{0};
    std::atomic<int64_t> pending_delta{0};
  };

  void accept_loop();
  void handle_connection(int fd);
  bool rl_check(double* retry_after_s, std::string* rule_name);
  void rl_charge(const Usage& u);
  void rl_roll(RuleState& rs, int64_t now_ms);

  std::shared_ptr<const std::vector<FastRoute>> routes_ =
      std::make_shared<const std::vector<FastRoute>>();
  std::vector<std::unique_ptr<RuleState>> rules_;
  std::string fallback_host_;
  uint16_t fallback_port_ = 0;
  std::vector<std::unique_ptr<GpuAdmissionClient>> gpu_;
  std::unique_ptr<DirectGpuBatcher> gpu_direct_;
  std::atomic<uint64_t> gpu_rr_{0};
  int64_t gpu_count(const std::string& text);
  bool gpu_enabled() const;
  std::unique_ptr<UpstreamPool> pool_;
  ServerStats stats_;

  int listen_fd_ = -1;
  std::atomic<bool> stopping_{false};
  std::thread acceptor_;
  std::mutex conn_mu_;
  std::set<int> conn_fds_;
  std::mutex cred_mu_;
  std::map<std::string, FileCredState> file_creds_;
  std::mutex cache_mu_;
  std::vector<std::string> cache_values_;  // row -> fp(8) + mode(1) + body
};

// -------- native bench harness (see fastpath.cpp) ---------------------------

class FastMock {
 public:
  int start(const std::string& host, const std::string& response);
  void stop();
  uint64_t requests() const;
  ~FastMock() { stop(); }

 private:
  void serve(int fd);
  std::string response_;
  int listen_fd_ = -1;
  std::thread acceptor_;
  std::atomic<uint64_t> served_{0};
};

struct LoadResult {
  double elapsed_s = 0;
  uint64_t completed = 0;
  uint64_t errors = 0;
  double p50_ms = 0, p99_ms = 0;
};

LoadResult run_load(const std::string& host, uint16_t port,
                    const std::string& path, const std::string& body,
                    int connections, int per_conn);
LoadResult run_load_pool(const std::string& host, uint16_t port,
                         const std::string& path,
                         const std::vector<std::string>& bodies,
                         int connections, int per_conn);

}  // namespace aigw_fast
