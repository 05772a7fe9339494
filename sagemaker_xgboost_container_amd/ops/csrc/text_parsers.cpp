// Native text-data parsers (host side of the data loader).
//
// The reference's training-data parsing happens inside libxgboost's C++
// text parsers (reached via xgb.DMatrix(uri), data_utils.py:309-361). This
// provides the equivalent native path for this framework: a multi-threaded
// libsvm parser producing CSR arrays directly, ~50x the pure-Python line
// parser on large files. (CSV goes through pandas' C tokenizer already.)

#include <ATen/Parallel.h>
#include <torch/extension.h>

#include <atomic>
#include <cstdlib>
#include <cstring>
#include <limits>
#include <thread>
#include <vector>

namespace {

struct Chunk {
  std::vector<float> values;
  std::vector<int64_t> indices;
  std::vector<int64_t> row_sizes;
  std::vector<float> labels;
  std::vector<float> weights;
  std::vector<int64_t> qids;
  bool any_weight = false;
  bool any_qid = false;
  int64_t max_index = -1;
};

// parse [begin, end) which is aligned to line boundaries
void parse_span(const char* begin, const char* end, Chunk& out) {
  const char* p = begin;
  while (p < end) {
    // skip leading whitespace / blank lines
    while (p < end && (*p == '\n' || *p == '\r' || *p == ' ' || *p == '\t')) ++p;
    if (p >= end) break;
    if (*p == '#') {  // comment line
      while (p < end && *p != '\n') ++p;
      continue;
    }
    char* next = nullptr;
    float label = std::strtof(p, &next);
    p = next;
    float weight = 1.0f;
    if (p < end && *p == ':') {  // label:weight extension
      ++p;
      weight = std::strtof(p, &next);
      p = next;
      out.any_weight = true;
    }
    int64_t nnz = 0;
    int64_t qid = 0;
    while (p < end && *p != '\n') {
      while (p < end && (*p == ' ' || *p == '\t' || *p == '\r')) ++p;
      if (p >= end || *p == '\n' || *p == '#') {
        while (p < end && *p != '\n') ++p;
        break;
      }
      if (p + 4 <= end && std::strncmp(p, "qid:", 4) == 0) {
        p += 4;
        qid = std::strtoll(p, &next, 10);
        out.any_qid = true;
        p = next;
        continue;
      }
      int64_t idx = std::strtoll(p, &next, 10);
      if (next == p || *next != ':') {  // malformed token: skip it
        while (p < end && *p != ' ' && *p != '\n') ++p;
        continue;
      }
      p = next + 1;
      float value = std::strtof(p, &next);
      p = next;
      out.indices.push_back(idx);
      out.values.push_back(value);
      if (idx > out.max_index) out.max_index = idx;
      ++nnz;
    }
    out.labels.push_back(label);
    out.weights.push_back(weight);
    out.qids.push_back(qid);
    out.row_sizes.push_back(nnz);
  }
}

}  // namespace

// Returns (values f32, indices i64, indptr i64, labels f32, weights f32 [0 if
// none], qids i64 [0 if none], num_col)
std::vector<torch::Tensor> parse_libsvm(const std::string& text, int64_t nthreads) {
  const char* data = text.data();
  const int64_t size = (int64_t)text.size();
  int64_t nt = nthreads > 0 ? nthreads : (int64_t)std::thread::hardware_concurrency();
  if (nt < 1) nt = 1;
  if (size < (1 << 16)) nt = 1;

  // chunk boundaries aligned to newlines
  std::vector<int64_t> bounds(nt + 1, size);
  bounds[0] = 0;
  for (int64_t t = 1; t < nt; ++t) {
    int64_t pos = size * t / nt;
    while (pos < size && data[pos] != '\n') ++pos;
    bounds[t] = pos < size ? pos + 1 : size;
  }

  std::vector<Chunk> chunks(nt);
  std::vector<std::thread> threads;
  for (int64_t t = 0; t < nt; ++t) {
    threads.emplace_back([&, t] { parse_span(data + bounds[t], data + bounds[t + 1], chunks[t]); });
  }
  for (auto& th : threads) th.join();

  int64_t total_rows = 0, total_nnz = 0, max_index = -1;
  bool any_weight = false, any_qid = false;
  for (auto& c : chunks) {
    total_rows += (int64_t)c.labels.size();
    total_nnz += (int64_t)c.values.size();
    max_index = std::max(max_index, c.max_index);
    any_weight |= c.any_weight;
    any_qid |= c.any_qid;
  }

  auto values = torch::empty({total_nnz}, torch::kFloat32);
  auto indices = torch::empty({total_nnz}, torch::kInt64);
  auto indptr = torch::empty({total_rows + 1}, torch::kInt64);
  auto labels = torch::empty({total_rows}, torch::kFloat32);
  auto weights = torch::empty({any_weight ? total_rows : 0}, torch::kFloat32);
  auto qids = torch::empty({any_qid ? total_rows : 0}, torch::kInt64);

  float* vp = values.data_ptr<float>();
  int64_t* ip = indices.data_ptr<int64_t>();
  int64_t* pp = indptr.data_ptr<int64_t>();
  float* lp = labels.data_ptr<float>();
  float* wp = any_weight ? weights.data_ptr<float>() : nullptr;
  int64_t* qp = any_qid ? qids.data_ptr<int64_t>() : nullptr;

  int64_t row = 0, nnz = 0;
  pp[0] = 0;
  for (auto& c : chunks) {
    std::memcpy(vp + nnz, c.values.data(), c.values.size() * sizeof(float));
    std::memcpy(ip + nnz, c.indices.data(), c.indices.size() * sizeof(int64_t));
    std::memcpy(lp + row, c.labels.data(), c.labels.size() * sizeof(float));
    if (wp) std::memcpy(wp + row, c.weights.data(), c.weights.size() * sizeof(float));
    if (qp) std::memcpy(qp + row, c.qids.data(), c.qids.size() * sizeof(int64_t));
    for (int64_t r = 0; r < (int64_t)c.row_sizes.size(); ++r) {
      nnz += c.row_sizes[r];
      pp[row + r + 1] = nnz;
    }
    row += (int64_t)c.labels.size();
  }

  auto ncol = torch::tensor({max_index + 1}, torch::kInt64);
  return {values, indices, indptr, labels, weights, qids, ncol};
}

// Batched forest traversal on host (serving without a GPU): parallel over
// rows via at::parallel_for (respects torch.set_num_threads / nthread HP).
void predict_forest_cpu(torch::Tensor X, torch::Tensor left, torch::Tensor right,
                        torch::Tensor feat, torch::Tensor thresh, torch::Tensor defl,
                        torch::Tensor value, torch::Tensor tree_root, torch::Tensor tree_cls,
                        int64_t t_begin, int64_t t_end, torch::Tensor out, int64_t k) {
  const float* x = X.data_ptr<float>();
  const int64_t n = X.size(0);
  const int64_t nf = X.size(1);
  const int* lp = left.data_ptr<int>();
  const int* rp = right.data_ptr<int>();
  const int* fp = feat.data_ptr<int>();
  const float* tp = thresh.data_ptr<float>();
  const unsigned char* dp = defl.data_ptr<unsigned char>();
  const float* vp = value.data_ptr<float>();
  const int* roots = tree_root.data_ptr<int>();
  const int* cls = tree_cls.data_ptr<int>();
  float* op = out.data_ptr<float>();

  at::parallel_for(0, n, 64, [&](int64_t lo, int64_t hi) {
    for (int64_t row = lo; row < hi; ++row) {
      const float* xr = x + row * nf;
      float* orow = op + row * k;
      for (int64_t t = t_begin; t < t_end; ++t) {
        int nid = roots[t];
        int l;
        while ((l = lp[nid]) >= 0) {
          const float fv = xr[fp[nid]];
          const bool goleft = std::isnan(fv) ? (dp[nid] != 0) : (fv < tp[nid]);
          nid = goleft ? l : rp[nid];
        }
        orow[cls[t]] += vp[nid];
      }
    }
  });
}

// Multi-threaded CSV parser: numeric fields, empty -> NaN, arbitrary
// single-char delimiter. Two passes: count rows/cols, then parse into a
// dense float32 tensor in parallel.
torch::Tensor parse_csv(const std::string& text, const std::string& delimiter, int64_t nthreads) {
  const char delim = delimiter.empty() ? ',' : delimiter[0];
  const char* data = text.data();
  const int64_t size = (int64_t)text.size();

  // column count from the first non-empty line
  int64_t ncol = 0;
  {
    int64_t p = 0;
    while (p < size && (data[p] == '\n' || data[p] == '\r')) ++p;
    int64_t cols = 1;
    while (p < size && data[p] != '\n') {
      if (data[p] == delim) ++cols;
      ++p;
    }
    ncol = cols;
  }
  // row count
  int64_t nrow = 0;
  {
    bool in_line = false;
    for (int64_t p = 0; p < size; ++p) {
      if (data[p] == '\n') {
        if (in_line) ++nrow;
        in_line = false;
      } else if (data[p] != '\r') {
        in_line = true;
      }
    }
    if (in_line) ++nrow;
  }

  auto out = torch::empty({nrow, ncol}, torch::kFloat32);
  float* op = out.data_ptr<float>();

  int64_t nt = nthreads > 0 ? nthreads : (int64_t)std::thread::hardware_concurrency();
  if (nt < 1) nt = 1;
  if (size < (1 << 16)) nt = 1;

  // chunk by line boundaries; each chunk counts its starting row first
  std::vector<int64_t> bounds(nt + 1, size);
  bounds[0] = 0;
  for (int64_t t = 1; t < nt; ++t) {
    int64_t p = size * t / nt;
    while (p < size && data[p] != '\n') ++p;
    bounds[t] = p < size ? p + 1 : size;
  }
  std::vector<int64_t> start_row(nt + 1, 0);
  {
    int64_t row = 0;
    int64_t t = 1;
    bool in_line = false;
    for (int64_t p = 0; p < size && t <= nt; ++p) {
      while (t <= nt && p == bounds[t]) start_row[t++] = row;
      if (data[p] == '\n') {
        if (in_line) ++row;
        in_line = false;
      } else if (data[p] != '\r') {
        in_line = true;
      }
    }
  }

  const float nan_v = std::numeric_limits<float>::quiet_NaN();
  std::vector<std::thread> threads;
  for (int64_t t = 0; t < nt; ++t) {
    threads.emplace_back([&, t] {
      const char* p = data + bounds[t];
      const char* end = data + bounds[t + 1];
      int64_t row = start_row[t];
      while (p < end) {
        while (p < end && (*p == '\n' || *p == '\r')) ++p;
        if (p >= end) break;
        float* rp = op + row * ncol;
        for (int64_t c = 0; c < ncol; ++c) {
          // empty field -> NaN
          if (p >= end || *p == delim || *p == '\n' || *p == '\r') {
            rp[c] = nan_v;
          } else {
            char* next = nullptr;
            rp[c] = std::strtof(p, &next);
            p = (next == p) ? p : next;
            while (p < end && *p != delim && *p != '\n') ++p;  // trailing junk
          }
          if (p < end && *p == delim) ++p;
        }
        while (p < end && *p != '\n') ++p;
        ++row;
      }
    });
  }
  for (auto& th : threads) th.join();
  return out;
}

void init_text_parsers(pybind11::module_& m) {
  m.def("parse_libsvm", &parse_libsvm, "multi-threaded libsvm -> CSR parser",
        pybind11::arg("text"), pybind11::arg("nthreads") = 0);
  m.def("predict_forest_cpu", &predict_forest_cpu, "parallel host forest traversal");
  m.def("parse_csv", &parse_csv, "multi-threaded csv -> dense float32 parser",
        pybind11::arg("text"), pybind11::arg("delimiter") = ",", pybind11::arg("nthreads") = 0);
}
