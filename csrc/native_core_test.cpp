// Standalone sanitizer test for the native hot-path cores (no Python).
// Built by scripts/sanitize_native.sh with -fsanitize=address,undefined
// and -D_GLIBCXX_ASSERTIONS — the memory-safety/race analogue of the
// reference's `go test -race` CI lane. Covers:
//  - scan_chat_body core: extraction correctness, malformed-input
//    rejection, and a random-bytes fuzz loop (ASan catches overreads);
//  - SSECore: chunk-boundary invariance (byte-at-a-time == one-shot)
//    over randomized chunkings of a canned stream.

#include <cassert>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "native_core.h"

using aigw_core::Scan;
using aigw_core::SSECore;

static int failures = 0;
#define CHECK(cond)                                                    \
  do {                                                                 \
    if (!(cond)) {                                                     \
      std::fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond); \
      ++failures;                                                      \
    }                                                                  \
  } while (0)

struct ScanResult {
  bool ok;
  std::string model;
  bool stream;
  std::string text;
};

static ScanResult scan(const std::string& body) {
  Scan sc{body.data(), body.data() + body.size()};
  bool ok = sc.parse_value(0, "", true) && sc.ok;
  if (ok) {
    sc.ws();
    ok = sc.p == sc.end;
  }
  return {ok, sc.model, ok && sc.stream == 1, sc.text};
}

static void test_scan_extraction() {
  auto r = scan(R"({"model":"gpt-4o","stream":true,"messages":[)"
                R"({"role":"user","content":"hi there"}]})");
  CHECK(r.ok && r.model == "gpt-4o" && r.stream);
  CHECK(r.text == "hi there\n");

  r = scan(R"({"system":"be nice","model":"m","messages":[{"content":)"
           R"([{"type":"text","text":"a"},{"type":"text","text":"b"}]}]})");
  CHECK(r.ok && r.text == "be nice\na\nb\n");

  // escapes incl. surrogate pair (clef = U+1D11E)
  r = scan("{\"model\":\"x\",\"messages\":[{\"content\":"
           "\"q\\n\\\"\\u0041\\uD834\\uDD1E\"}]}");
  CHECK(r.ok);
  CHECK(r.text == std::string("q\n\"A\xF0\x9D\x84\x9E\n"));

  // text/content OUTSIDE the messages tree is never collected (keeps
  // the fast path identical to the strict python extractor)
  r = scan(R"({"model":"m","metadata":{"text":"noise","content":"x"},)"
           R"("messages":[{"content":"real"}]})");
  CHECK(r.ok && r.text == "real\n");

  // stream:false and absent stream
  CHECK(!scan(R"({"model":"m","stream":false})").stream);
  CHECK(!scan(R"({"model":"m"})").stream);
  // nested "model" key must NOT win (depth-1 rule)
  r = scan(R"({"a":{"model":"inner"},"model":"outer"})");
  CHECK(r.ok && r.model == "outer");
}

static void test_scan_rejects() {
  CHECK(!scan("").ok);
  CHECK(!scan("{").ok);
  CHECK(!scan(R"({"a":1)").ok);
  CHECK(!scan(R"({"a":})").ok);
  CHECK(!scan(R"({"a":"unterminated)").ok);
  CHECK(!scan(R"({"a":"bad\q"})").ok);
  CHECK(!scan(R"({"a":"\u12"})").ok);
  CHECK(!scan(R"({"a":1} trailing)").ok);
  CHECK(!scan(R"([1,2,)").ok);
  std::string deep;  // depth bomb: 80 > the 64 depth cap
  for (int i = 0; i < 80; ++i) deep += "[";
  CHECK(!scan(deep).ok);
}

static uint64_t rng_state = 0x2545F4914F6CDD1DULL;
static uint32_t next_rand() {
  rng_state ^= rng_state << 13;
  rng_state ^= rng_state >> 7;
  rng_state ^= rng_state << 17;
  return (uint32_t)(rng_state >> 32);
}

static void fuzz_scan_random_bytes() {
  // random byte soup: must never crash/overread (ASan enforces), and a
  // successful parse must consume the whole input.
  const char alphabet[] = "{}[]\",:\\utrue falsn0123456789.eE-+\x01\xff\x80 ";
  for (int iter = 0; iter < 20000; ++iter) {
    size_t len = next_rand() % 64;
    std::string body;
    for (size_t i = 0; i < len; ++i)
      body.push_back(alphabet[next_rand() % (sizeof(alphabet) - 1)]);
    (void)scan(body);
  }
  // mutations of a valid body
  const std::string valid =
      R"({"model":"gpt-4o","stream":true,"messages":[{"content":"hello é"}]})";
  for (int iter = 0; iter < 20000; ++iter) {
    std::string body = valid;
    size_t pos = next_rand() % body.size();
    body[pos] = (char)(next_rand() % 256);
    (void)scan(body);
  }
}

using Events = std::vector<std::pair<std::string, std::string>>;

static Events run_sse(const std::string& stream, const std::vector<size_t>& cuts) {
  SSECore core;
  Events out;
  auto emit = [&out](const std::string& ev, const std::string& data) {
    out.emplace_back(ev, data);
  };
  size_t pos = 0;
  for (size_t cut : cuts) {
    size_t n = std::min(cut, stream.size() - pos);
    core.feed(stream.data() + pos, n, emit);
    pos += n;
    if (pos >= stream.size()) break;
  }
  if (pos < stream.size()) core.feed(stream.data() + pos, stream.size() - pos, emit);
  core.flush(emit);
  return out;
}

static void test_sse_semantics() {
  const std::string s =
      ": comment\n"
      "event: delta\n"
      "data: {\"a\":1}\n"
      "\n"
      "data:\n"
      "data:\n"
      "\n"
      "data: [DONE]\r\n"
      "\r\n"
      "data: tail-no-blank";
  Events e = run_sse(s, {s.size()});
  CHECK(e.size() == 4);
  CHECK(e[0].first == "delta" && e[0].second == "{\"a\":1}");
  CHECK(e[1].first.empty() && e[1].second == "\n");  // two empty data lines join
  CHECK(e[2].second == "[DONE]");
  CHECK(e[3].second == "tail-no-blank");  // delivered by flush
}

static void test_sse_chunk_invariance() {
  const std::string s =
      "event: one\ndata: aaa\ndata: bbb\n\n"
      ": keepalive\r\n"
      "data: {\"x\": \"y\"}\n\nid: 7\ndata: last\n\n";
  Events whole = run_sse(s, {s.size()});
  CHECK(whole.size() == 3);
  CHECK(whole[0].second == "aaa\nbbb");
  // byte-at-a-time
  std::vector<size_t> bytes(s.size(), 1);
  CHECK(run_sse(s, bytes) == whole);
  // random chunkings
  for (int iter = 0; iter < 5000; ++iter) {
    std::vector<size_t> cuts;
    size_t total = 0;
    while (total < s.size()) {
      size_t c = 1 + next_rand() % 7;
      cuts.push_back(c);
      total += c;
    }
    Events got = run_sse(s, cuts);
    CHECK(got == whole);
    if (got != whole) return;  // don't spam
  }
}

int main() {
  test_scan_extraction();
  test_scan_rejects();
  fuzz_scan_random_bytes();
  test_sse_semantics();
  test_sse_chunk_invariance();
  if (failures) {
    std::fprintf(stderr, "%d failure(s)\n", failures);
    return 1;
  }
  std::printf("native core sanitizer tests passed\n");
  return 0;
}
