// pybind11 bindings for the native fast-path server (csrc/fastpath.cpp).
// CPython composes the config (from aigw.filterapi.RuntimeConfig) and
// owns the cold-path fallback app; the server itself runs entirely in
// native threads with the GIL released.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "fastpath.h"

namespace py = pybind11;
using namespace aigw_fast;

PYBIND11_MODULE(aigw_fast, m) {
  m.doc() = "aigw native data-plane fast path";

  py::class_<FastMock>(m, "FastMock")
      .def(py::init<>())
      .def("start", &FastMock::start, py::arg("host"), py::arg("response"))
      .def("stop", &FastMock::stop)
      .def("requests", &FastMock::requests);

  m.def(
      "run_load",
      [](const std::string& host, uint16_t port, const std::string& path,
         py::bytes body, int connections, int per_conn) {
        LoadResult r;
        {
          std::string b = body;
          py::gil_scoped_release release;
          r = run_load(host, port, path, b, connections, per_conn);
        }
        py::dict d;
        d["elapsed_s"] = r.elapsed_s;
        d["completed"] = r.completed;
        d["errors"] = r.errors;
        d["p50_ms"] = r.p50_ms;
        d["p99_ms"] = r.p99_ms;
        return d;
      },
      py::arg("host"), py::arg("port"), py::arg("path"), py::arg("body"),
      py::arg("connections"), py::arg("per_conn"),
      "closed-loop native load generator (threads, keep-alive)");

  m.def(
      "run_load_pool",
      [](const std::string& host, uint16_t port, const std::string& path,
         py::list bodies, int connections, int per_conn) {
        std::vector<std::string> pool;
        for (auto b : bodies) pool.push_back(b.cast<py::bytes>());
        LoadResult r;
        {
          py::gil_scoped_release release;
          r = run_load_pool(host, port, path, pool, connections, per_conn);
        }
        py::dict d;
        d["elapsed_s"] = r.elapsed_s;
        d["completed"] = r.completed;
        d["errors"] = r.errors;
        d["p50_ms"] = r.p50_ms;
        d["p99_ms"] = r.p99_ms;
        return d;
      },
      py::arg("host"), py::arg("port"), py::arg("path"), py::arg("bodies"),
      py::arg("connections"), py::arg("per_conn"),
      "run_load with a cycling body pool (semantic-cache hit mixes)");

  py::class_<FastServer>(m, "FastServer")
      .def(py::init<>())
      .def(
          "add_route",
          [](FastServer& s, const std::string& name, const std::string& model_match,
             int retries, bool has_costs, bool eligible, py::list backends) {
            FastRoute r;
            r.name = name;
            r.model_match = model_match;
            r.retries = retries;
            r.has_costs = has_costs;
            r.eligible = eligible;
            for (auto item : backends) {
              py::dict d = item.cast<py::dict>();
              FastBackend b;
              b.name = d["name"].cast<std::string>();
              b.host = d["host"].cast<std::string>();
              b.port = d["port"].cast<uint16_t>();
              if (d.contains("bearer")) b.bearer = d["bearer"].cast<std::string>();
              if (d.contains("api_key_file"))
                b.api_key_file = d["api_key_file"].cast<std::string>();
              if (d.contains("model_override"))
                b.model_override = d["model_override"].cast<std::string>();
              if (d.contains("weight")) b.weight = d["weight"].cast<double>();
              if (d.contains("priority")) b.priority = d["priority"].cast<int>();
              if (d.contains("timeout_s")) b.timeout_s = d["timeout_s"].cast<double>();
              if (d.contains("azure")) b.azure = d["azure"].cast<bool>();
              if (d.contains("azure_api_version"))
                b.azure_api_version = d["azure_api_version"].cast<std::string>();
              r.backends.push_back(std::move(b));
            }
            s.add_route(std::move(r));
          },
          py::arg("name"), py::arg("model_match"), py::arg("retries"),
          py::arg("has_costs"), py::arg("eligible"), py::arg("backends"))
      .def("add_rate_rule",
           [](FastServer& s, const std::string& name, int64_t limit,
              double window_s, const std::string& metadata_key) {
             RateRule r;
             r.name = name;
             r.limit = limit;
             r.window_s = window_s;
             r.metadata_key = metadata_key == "llm_input_token"    ? 1
                              : metadata_key == "llm_output_token" ? 2
                                                                   : 0;
             s.add_rate_rule(r);
           })
      .def("set_fallback", &FastServer::set_fallback)
      .def(
          "swap_routes",
          [](FastServer& s, py::list route_specs) {
            std::vector<FastRoute> routes;
            for (auto item : route_specs) {
              py::dict rd = item.cast<py::dict>();
              FastRoute r;
              r.name = rd["name"].cast<std::string>();
              r.model_match = rd["model_match"].cast<std::string>();
              r.retries = rd["retries"].cast<int>();
              r.has_costs = rd["has_costs"].cast<bool>();
              r.eligible = rd["eligible"].cast<bool>();
              for (auto b : rd["backends"].cast<py::list>()) {
                py::dict d = b.cast<py::dict>();
                FastBackend be;
                be.name = d["name"].cast<std::string>();
                be.host = d["host"].cast<std::string>();
                be.port = d["port"].cast<uint16_t>();
                if (d.contains("bearer")) be.bearer = d["bearer"].cast<std::string>();
                if (d.contains("api_key_file"))
                  be.api_key_file = d["api_key_file"].cast<std::string>();
                if (d.contains("model_override"))
                  be.model_override = d["model_override"].cast<std::string>();
                if (d.contains("weight")) be.weight = d["weight"].cast<double>();
                if (d.contains("priority")) be.priority = d["priority"].cast<int>();
                if (d.contains("timeout_s")) be.timeout_s = d["timeout_s"].cast<double>();
                if (d.contains("azure")) be.azure = d["azure"].cast<bool>();
                if (d.contains("azure_api_version"))
                  be.azure_api_version = d["azure_api_version"].cast<std::string>();
                r.backends.push_back(std::move(be));
              }
              routes.push_back(std::move(r));
            }
            s.swap_routes(std::move(routes));
          },
          "hot-swap the route table (in-flight requests keep the old one)")
      .def("drain", &FastServer::drain, py::arg("drain_s"),
           py::call_guard<py::gil_scoped_release>())
      .def("enable_gpu", &FastServer::enable_gpu, py::arg("socket_path"),
           py::arg("window_us") = 100, py::arg("max_batch") = 256)
      .def(
          "enable_gpu_direct",
          [](FastServer& s, py::buffer htab_keys, py::buffer htab_rank,
             int max_batch, size_t max_batch_bytes, int max_req, int device) {
            py::buffer_info ki = htab_keys.request();
            py::buffer_info ri = htab_rank.request();
            if (ki.itemsize != 8 || ri.itemsize != 4)
              throw std::runtime_error(
                  "htab_keys must be int64, htab_rank int32");
            if (ki.size != ri.size)
              throw std::runtime_error("htab arrays must have equal length");
            s.enable_gpu_direct(static_cast<const long long*>(ki.ptr),
                                static_cast<const int32_t*>(ri.ptr),
                                (int)ki.size, max_batch, max_batch_bytes,
                                max_req, device);
          },
          py::arg("htab_keys"), py::arg("htab_rank"),
          py::arg("max_batch") = 1024,
          py::arg("max_batch_bytes") = (size_t)48 * 1024 * 1024,
          py::arg("max_req") = 4096, py::arg("device") = 0)
      .def(
          "enable_gpu_direct_cache",
          [](FastServer& s, py::buffer emb, py::buffer proj, int dim,
             long long capacity, float threshold, bool fp8) {
            py::buffer_info ei = emb.request();
            py::buffer_info pi = proj.request();
            if (ei.itemsize != 2 || pi.itemsize != 2)
              throw std::runtime_error("emb/proj must be bf16 (uint16 view)");
            int vocab = (int)(ei.size / dim);
            if ((long long)vocab * dim != ei.size || pi.size != (long long)dim * dim)
              throw std::runtime_error("emb/proj shape mismatch");
            if (!s.enable_gpu_direct_cache(
                    static_cast<const uint16_t*>(ei.ptr), vocab,
                    static_cast<const uint16_t*>(pi.ptr), dim, capacity,
                    threshold, fp8))
              throw std::runtime_error("native cache init failed");
          },
          py::arg("emb"), py::arg("proj"), py::arg("dim") = 384,
          py::arg("capacity") = 65536, py::arg("threshold") = 0.92,
          py::arg("fp8") = false)
      .def("start", &FastServer::start, py::arg("host"), py::arg("port"),
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &FastServer::stop, py::call_guard<py::gil_scoped_release>())
      .def("gpu_direct_stats", &FastServer::gpu_direct_stats)
      .def("rl_collect_deltas", &FastServer::rl_collect_deltas)
      .def("rl_apply_remote", &FastServer::rl_apply_remote)
      .def("rl_local_spent", &FastServer::rl_local_spent)
      .def("stats", [](const FastServer& s) {
        const ServerStats& st = s.stats();
        py::dict d;
        d["requests"] = st.requests.load();
        d["responses_2xx"] = st.responses_2xx.load();
        d["responses_4xx"] = st.responses_4xx.load();
        d["responses_5xx"] = st.responses_5xx.load();
        d["local_429"] = st.local_429.load();
        d["fallback"] = st.fallback.load();
        d["retries"] = st.retries.load();
        d["gpu_tokens"] = st.gpu_tokens.load();
        d["cache_hits"] = st.cache_hits.load();
        d["cache_misses"] = st.cache_misses.load();
        d["input_tokens"] = st.input_tokens.load();
        d["output_tokens"] = st.output_tokens.load();
        d["total_tokens"] = st.total_tokens.load();
        d["bytes_in"] = st.bytes_in.load();
        d["bytes_out"] = st.bytes_out.load();
        d["active_connections"] = st.active_connections.load();
        py::list hist;
        for (int i = 0; i < 32; ++i) hist.append(st.latency_us_log2[i].load());
        d["latency_us_log2"] = hist;
        return d;
      });
}
