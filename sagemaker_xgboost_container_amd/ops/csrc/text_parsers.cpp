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
  std::atomic<int64_t> bad_fields{0};
  for (int64_t t = 0; t < nt; ++t) {
    threads.emplace_back([&, t] {
      const char* p = data + bounds[t];
      const char* end = data + bounds[t + 1];
      int64_t row = start_row[t];
      int64_t bad = 0;
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
            if (next == p) ++bad;  // non-empty, non-numeric field
            p = (next == p) ? p : next;
            while (p < end && *p != delim && *p != '\n') ++p;  // trailing junk
          }
          if (p < end && *p == delim) ++p;
        }
        while (p < end && *p != '\n') ++p;
        ++row;
      }
      bad_fields += bad;
    });
  }
  for (auto& th : threads) th.join();
  if (bad_fields.load() > 0) {
    throw std::runtime_error(
        "could not convert " + std::to_string(bad_fields.load()) +
        " csv field(s) to float (non-numeric data)");
  }
  return out;
}

// ---------------------------------------------------------------------------
// Exact TreeSHAP (Lundberg et al., Algorithm 2) — replaces the reference's
// native pred_contribs path (booster.predict(pred_contribs=True),
// reference test/integration/local/test_abalone.py:65). Parallel over rows
// with at::parallel_for; per-row recursion over each tree with the
// polynomial-time path-weight bookkeeping.
// ---------------------------------------------------------------------------
namespace treeshap {

struct PathElement {
  int feature_index;
  double zero_fraction;  // proportion of "cold" paths flowing through
  double one_fraction;   // 1 when x follows this split, else 0
  double pweight;        // permutation weight accumulated so far
};

inline void extend_path(PathElement* path, int unique_depth, double zero_fraction,
                        double one_fraction, int feature_index) {
  path[unique_depth].feature_index = feature_index;
  path[unique_depth].zero_fraction = zero_fraction;
  path[unique_depth].one_fraction = one_fraction;
  path[unique_depth].pweight = unique_depth == 0 ? 1.0 : 0.0;
  const double inv = 1.0 / (unique_depth + 1);
  for (int i = unique_depth - 1; i >= 0; --i) {
    path[i + 1].pweight += one_fraction * path[i].pweight * (i + 1) * inv;
    path[i].pweight = zero_fraction * path[i].pweight * (unique_depth - i) * inv;
  }
}

inline void unwind_path(PathElement* path, int unique_depth, int path_index) {
  const double one_fraction = path[path_index].one_fraction;
  const double zero_fraction = path[path_index].zero_fraction;
  double next_one_portion = path[unique_depth].pweight;
  for (int i = unique_depth - 1; i >= 0; --i) {
    if (one_fraction != 0.0) {
      const double tmp = path[i].pweight;
      path[i].pweight = next_one_portion * (unique_depth + 1) / ((i + 1) * one_fraction);
      next_one_portion = tmp - path[i].pweight * zero_fraction * (unique_depth - i) /
                                   (double)(unique_depth + 1);
    } else {
      path[i].pweight = path[i].pweight * (unique_depth + 1) /
                        (zero_fraction * (unique_depth - i));
    }
  }
  for (int i = path_index; i < unique_depth; ++i) {
    path[i].feature_index = path[i + 1].feature_index;
    path[i].zero_fraction = path[i + 1].zero_fraction;
    path[i].one_fraction = path[i + 1].one_fraction;
  }
}

inline double unwound_path_sum(const PathElement* path, int unique_depth, int path_index) {
  const double one_fraction = path[path_index].one_fraction;
  const double zero_fraction = path[path_index].zero_fraction;
  double next_one_portion = path[unique_depth].pweight;
  double total = 0.0;
  for (int i = unique_depth - 1; i >= 0; --i) {
    if (one_fraction != 0.0) {
      const double tmp = next_one_portion * (unique_depth + 1) / ((i + 1) * one_fraction);
      total += tmp;
      next_one_portion = path[i].pweight - tmp * zero_fraction * (unique_depth - i) /
                                               (double)(unique_depth + 1);
    } else if (zero_fraction != 0.0) {
      total += (path[i].pweight / zero_fraction) * (unique_depth + 1) /
               (double)(unique_depth - i);
    }
  }
  return total;
}

struct TreeView {
  const int* left;
  const int* right;
  const int* feat;
  const float* thresh;
  const unsigned char* defl;
  const float* value;
  const float* cover;
};

// recursion: `path` region for this call starts after the parent's copy
void recurse(const TreeView& t, const float* xr, double* phi, int node,
             int unique_depth, PathElement* parent_path, double parent_zero_fraction,
             double parent_one_fraction, int parent_feature_index) {
  PathElement* path = parent_path + unique_depth + 1;
  for (int i = 0; i < unique_depth; ++i) path[i] = parent_path[i];
  extend_path(path, unique_depth, parent_zero_fraction, parent_one_fraction,
              parent_feature_index);

  const int l = t.left[node];
  if (l < 0) {  // leaf
    const double v = t.value[node];
    for (int i = 1; i <= unique_depth; ++i) {
      const double w = unwound_path_sum(path, unique_depth, i);
      phi[path[i].feature_index] +=
          w * (path[i].one_fraction - path[i].zero_fraction) * v;
    }
    return;
  }

  const int r = t.right[node];
  const int split = t.feat[node];
  const float fv = xr[split];
  const int hot = std::isnan(fv) ? (t.defl[node] ? l : r) : (fv < t.thresh[node] ? l : r);
  const int cold = hot == l ? r : l;
  const double node_cover = t.cover[node] > 0 ? t.cover[node] : 1.0;
  const double hot_zero_fraction = t.cover[hot] / node_cover;
  const double cold_zero_fraction = t.cover[cold] / node_cover;
  double incoming_zero_fraction = 1.0;
  double incoming_one_fraction = 1.0;

  // a previous split on the same feature is undone before extending
  int path_index = 0;
  for (; path_index <= unique_depth; ++path_index) {
    if (path[path_index].feature_index == split) break;
  }
  if (path_index != unique_depth + 1) {
    incoming_zero_fraction = path[path_index].zero_fraction;
    incoming_one_fraction = path[path_index].one_fraction;
    unwind_path(path, unique_depth, path_index);
    unique_depth -= 1;
  }

  recurse(t, xr, phi, hot, unique_depth + 1, path,
          hot_zero_fraction * incoming_zero_fraction, incoming_one_fraction, split);
  recurse(t, xr, phi, cold, unique_depth + 1, path,
          cold_zero_fraction * incoming_zero_fraction, 0.0, split);
}

}  // namespace treeshap

// out_phi: (n, k, f+1) float64, zero-init (bias column filled by the caller).
void tree_shap_cpu(torch::Tensor X, torch::Tensor left, torch::Tensor right,
                   torch::Tensor feat, torch::Tensor thresh, torch::Tensor defl,
                   torch::Tensor value, torch::Tensor cover, torch::Tensor tree_root,
                   torch::Tensor tree_cls, int64_t t_begin, int64_t t_end,
                   torch::Tensor out_phi, int64_t k, int64_t max_depth) {
  const float* x = X.data_ptr<float>();
  const int64_t n = X.size(0);
  const int64_t nf = X.size(1);
  const int64_t ncols = out_phi.size(2);  // nf + 1
  treeshap::TreeView tv{left.data_ptr<int>(),  right.data_ptr<int>(),
                        feat.data_ptr<int>(),  thresh.data_ptr<float>(),
                        defl.data_ptr<unsigned char>(), value.data_ptr<float>(),
                        cover.data_ptr<float>()};
  const int* roots = tree_root.data_ptr<int>();
  const int* cls = tree_cls.data_ptr<int>();
  double* op = out_phi.data_ptr<double>();
  // path scratch: call at depth d copies d elements and extends by one;
  // total region = sum_{d=0..D}(d+2) < (D+2)*(D+3)
  const int64_t scratch = (max_depth + 2) * (max_depth + 3);

  at::parallel_for(0, n, 16, [&](int64_t lo, int64_t hi) {
    std::vector<treeshap::PathElement> path((size_t)scratch);
    for (int64_t row = lo; row < hi; ++row) {
      const float* xr = x + row * nf;
      for (int64_t t = t_begin; t < t_end; ++t) {
        double* phi = op + (row * k + cls[t]) * ncols;
        treeshap::recurse(tv, xr, phi, roots[t], 0, path.data(), 1.0, 1.0, -1);
      }
    }
  });
}

// leaf index per (row, tree): vectorized pred_leaf (n, T) int32
void pred_leaf_cpu(torch::Tensor X, torch::Tensor left, torch::Tensor right,
                   torch::Tensor feat, torch::Tensor thresh, torch::Tensor defl,
                   torch::Tensor tree_root, int64_t t_begin, int64_t t_end,
                   torch::Tensor out) {
  const float* x = X.data_ptr<float>();
  const int64_t n = X.size(0);
  const int64_t nf = X.size(1);
  const int* lp = left.data_ptr<int>();
  const int* rp = right.data_ptr<int>();
  const int* fp = feat.data_ptr<int>();
  const float* tp = thresh.data_ptr<float>();
  const unsigned char* dp = defl.data_ptr<unsigned char>();
  const int* roots = tree_root.data_ptr<int>();
  const int64_t T = t_end - t_begin;
  int* op = out.data_ptr<int>();

  at::parallel_for(0, n, 64, [&](int64_t lo, int64_t hi) {
    for (int64_t row = lo; row < hi; ++row) {
      const float* xr = x + row * nf;
      int* orow = op + row * T;
      for (int64_t t = t_begin; t < t_end; ++t) {
        const int root = roots[t];
        int nid = root;
        int l;
        while ((l = lp[nid]) >= 0) {
          const float fv = xr[fp[nid]];
          const bool goleft = std::isnan(fv) ? (dp[nid] != 0) : (fv < tp[nid]);
          nid = goleft ? l : rp[nid];
        }
        orow[t - t_begin] = nid - root;  // tree-local leaf id (xgboost semantics)
      }
    }
  });
}

void init_text_parsers(pybind11::module_& m) {
  m.def("parse_libsvm", &parse_libsvm, "multi-threaded libsvm -> CSR parser",
        pybind11::arg("text"), pybind11::arg("nthreads") = 0);
  m.def("predict_forest_cpu", &predict_forest_cpu, "parallel host forest traversal");
  m.def("parse_csv", &parse_csv, "multi-threaded csv -> dense float32 parser",
        pybind11::arg("text"), pybind11::arg("delimiter") = ",", pybind11::arg("nthreads") = 0);
  m.def("tree_shap_cpu", &tree_shap_cpu, "exact TreeSHAP contributions (parallel rows)");
  m.def("pred_leaf_cpu", &pred_leaf_cpu, "parallel leaf-index traversal");
}
