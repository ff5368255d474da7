// Native text-format parsers for the data loader (CPU, multi-threaded).
// Reference: common/dataloader (HdfsDataSet record iteration) + the per-app
// parsers (NMFETDataParser.java:39, MLR/GBT/Lasso libsvm-style rows, LDA
// word lists). Python line-splitting is ~50x slower on the multi-GB inputs
// the reference targets; this scans the raw bytes of a split once per
// thread and emits torch tensors directly.

#include <torch/extension.h>

#include <atomic>
#include <cctype>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <thread>
#include <vector>

namespace {

struct Cursor {
  const char* p;
  const char* end;

  bool done() const { return p >= end; }
  void skip_ws_inline() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\r')) ++p;
  }
  void next_line() {
    while (p < end && *p != '\n') ++p;
    if (p < end) ++p;
  }
  bool at_comment_or_empty() {
    skip_ws_inline();
    return p >= end || *p == '#' || *p == '\n';
  }
  long read_long() {
    skip_ws_inline();
    char* q;
    long v = strtol(p, &q, 10);
    p = (q == p && p < end) ? p + 1 : q;   // always progress on garbage
    return v;
  }
  double read_double() {
    skip_ws_inline();
    char* q;
    double v = strtod(p, &q);
    p = (q == p && p < end) ? p + 1 : q;
    return v;
  }
  bool eol() {
    skip_ws_inline();
    return p >= end || *p == '\n';
  }
};

// split [start,end) into n chunks aligned to line starts
std::vector<std::pair<const char*, const char*>> line_chunks(
    const char* base, int64_t len, int n) {
  std::vector<std::pair<const char*, const char*>> out;
  std::vector<const char*> bounds(n + 1);
  bounds[0] = base;
  bounds[n] = base + len;
  for (int i = 1; i < n; ++i) {
    const char* p = base + (len * i) / n;
    while (p < base + len && *p != '\n') ++p;
    if (p < base + len) ++p;
    bounds[i] = p;
  }
  for (int i = 0; i < n; ++i)
    out.emplace_back(bounds[i], std::max(bounds[i], bounds[i + 1]));
  return out;
}

int nthreads() {
  // HARMONY_PARSE_THREADS caps the per-process scan threads: with W
  // loader processes per node (one per GPU), W * hw threads oversubscribe
  // the cores 8x and parsing runs 5x SLOWER than linear (measured in
  // scripts/ingest_bench.py) — the loader sets it to cores/world.
  if (const char* e = std::getenv("HARMONY_PARSE_THREADS")) {
    int v = std::atoi(e);
    if (v > 0) return std::min(v, 64);
  }
  unsigned hw = std::thread::hardware_concurrency();
  return std::max(1u, std::min(hw, 16u));
}

}  // namespace

// "rowId: col,val col,val ..." -> (rows, cols, vals)
std::vector<torch::Tensor> parse_nmf_bytes(const std::string& buf) {
  const int T = nthreads();
  auto chunks = line_chunks(buf.data(), (int64_t)buf.size(), T);
  std::vector<std::vector<int64_t>> rows(T), cols(T);
  std::vector<std::vector<float>> vals(T);
  std::vector<std::thread> ths;
  for (int t = 0; t < T; ++t) {
    ths.emplace_back([&, t] {
      Cursor c{chunks[t].first, chunks[t].second};
      while (!c.done()) {
        if (c.at_comment_or_empty()) { c.next_line(); continue; }
        long r = c.read_long();
        c.skip_ws_inline();
        if (c.p < c.end && *c.p == ':') ++c.p;
        while (!c.eol()) {
          long col = c.read_long();
          c.skip_ws_inline();
          if (c.p < c.end && *c.p == ',') ++c.p;
          double v = c.read_double();
          rows[t].push_back(r);
          cols[t].push_back(col);
          vals[t].push_back((float)v);
        }
        c.next_line();
      }
    });
  }
  for (auto& th : ths) th.join();
  int64_t total = 0;
  for (auto& v : rows) total += (int64_t)v.size();
  auto ro = torch::empty({total}, torch::kInt64);
  auto co = torch::empty({total}, torch::kInt64);
  auto vo = torch::empty({total}, torch::kFloat32);
  int64_t off = 0;
  for (int t = 0; t < T; ++t) {
    int64_t n = (int64_t)rows[t].size();
    if (n) {
      std::memcpy(ro.data_ptr<int64_t>() + off, rows[t].data(), n * 8);
      std::memcpy(co.data_ptr<int64_t>() + off, cols[t].data(), n * 8);
      std::memcpy(vo.data_ptr<float>() + off, vals[t].data(), n * 4);
    }
    off += n;
  }
  return {ro, co, vo};
}

// "label idx:val idx:val ..." -> (X dense [n,F], y [n])
std::vector<torch::Tensor> parse_libsvm_bytes(const std::string& buf,
                                              int64_t num_features) {
  const int T = nthreads();
  auto chunks = line_chunks(buf.data(), (int64_t)buf.size(), T);
  std::vector<std::vector<float>> Xs(T);
  std::vector<std::vector<float>> ys(T);
  std::vector<std::thread> ths;
  for (int t = 0; t < T; ++t) {
    ths.emplace_back([&, t] {
      Cursor c{chunks[t].first, chunks[t].second};
      while (!c.done()) {
        if (c.at_comment_or_empty()) { c.next_line(); continue; }
        double lab = c.read_double();
        size_t base = Xs[t].size();
        Xs[t].resize(base + num_features, 0.f);
        while (!c.eol()) {
          long idx = c.read_long();
          c.skip_ws_inline();
          if (c.p < c.end && *c.p == ':') ++c.p;
          double v = c.read_double();
          if (idx >= 0 && idx < num_features) Xs[t][base + idx] = (float)v;
        }
        ys[t].push_back((float)lab);
        c.next_line();
      }
    });
  }
  for (auto& th : ths) th.join();
  int64_t total = 0;
  for (auto& v : ys) total += (int64_t)v.size();
  auto X = torch::empty({total, num_features}, torch::kFloat32);
  auto y = torch::empty({total}, torch::kFloat32);
  int64_t off = 0;
  for (int t = 0; t < T; ++t) {
    int64_t n = (int64_t)ys[t].size();
    if (n) {
      std::memcpy(X.data_ptr<float>() + off * num_features, Xs[t].data(),
                  n * num_features * 4);
      std::memcpy(y.data_ptr<float>() + off, ys[t].data(), n * 4);
    }
    off += n;
  }
  return {X, y};
}

// "word word word ..." one doc per line -> (doc_offsets [n+1], words)
std::vector<torch::Tensor> parse_lda_bytes(const std::string& buf) {
  const int T = nthreads();
  auto chunks = line_chunks(buf.data(), (int64_t)buf.size(), T);
  std::vector<std::vector<int64_t>> words(T), lens(T);
  std::vector<std::thread> ths;
  for (int t = 0; t < T; ++t) {
    ths.emplace_back([&, t] {
      Cursor c{chunks[t].first, chunks[t].second};
      while (!c.done()) {
        if (c.at_comment_or_empty()) { c.next_line(); continue; }
        int64_t n = 0;
        while (!c.eol()) {
          words[t].push_back(c.read_long());
          ++n;
        }
        lens[t].push_back(n);
        c.next_line();
      }
    });
  }
  for (auto& th : ths) th.join();
  int64_t ndocs = 0, nwords = 0;
  for (int t = 0; t < T; ++t) {
    ndocs += (int64_t)lens[t].size();
    nwords += (int64_t)words[t].size();
  }
  auto off = torch::empty({ndocs + 1}, torch::kInt64);
  auto wo = torch::empty({nwords}, torch::kInt64);
  int64_t d = 0, w = 0;
  auto* op = off.data_ptr<int64_t>();
  op[0] = 0;
  for (int t = 0; t < T; ++t) {
    if (!words[t].empty())
      std::memcpy(wo.data_ptr<int64_t>() + w, words[t].data(),
                  words[t].size() * 8);
    w += (int64_t)words[t].size();
    for (int64_t L : lens[t]) {
      op[d + 1] = op[d] + L;
      ++d;
    }
  }
  return {off, wo};
}
