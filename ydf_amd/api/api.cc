// ydf_amd C++ user API implementation — see ydf_amd_api.h.
//
// Training drives the same C++ ops that twin the HIP kernels
// (../ops/cc/cpu_ops.cpp); model IO speaks the reference wire format
// directly (protobuf varint/length-delimited encoding emitted by hand,
// field numbers documented against the reference .proto sources —
// model/model_library.cc:92-107 directory layout,
// utils/blob_sequence.h:125-150 node shards,
// model/decision_tree/decision_tree.proto:202 Node records).
#include "ydf_amd_api.h"

#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstring>
#include <fstream>
#include <limits>
#include <map>
#include <set>
#include <functional>
#include <sstream>
#include <stdexcept>

#include "../ops/cc/common.h"

namespace ydfa {
// cpu_ops.cpp entry points (C linkage, same prototypes the pybind
// layer uses).
extern "C" {
void cpu_bin_data(const float*, const float*, uint8_t*, int64_t, int,
                  int, int);
void cpu_grad_hess(const float*, const float*, float*, int64_t, int);
void cpu_hist_build(const uint8_t*, const float*, const int32_t*,
                    const int32_t*, float*, int64_t, int, int, int, int,
                    int, int);
void cpu_split_scan(const float*, const int32_t*, float*, float*,
                    int32_t*, int32_t*, int32_t*, float*, const uint8_t*,
                    const uint8_t*, unsigned long long*, const int8_t*,
                    float*, int32_t*, uint8_t*, int, int, int, int,
                    SplitParams);
void cpu_update_node_ids(const uint8_t*, int32_t*, const int32_t*,
                         const int32_t*, const int32_t*, const uint8_t*,
                         const unsigned long long*, const uint8_t*,
                         int64_t, int, int);
void cpu_leaf_values(const float*, const float*, float*, int, float,
                     float);
void cpu_update_preds(float*, const int32_t*, const float*, int64_t,
                      float);
}  // extern "C"

namespace api {
namespace {

// ---------------------------------------------------------------------
// Protobuf wire primitives
// ---------------------------------------------------------------------
void put_varint(std::string& out, uint64_t v) {
  while (true) {
    uint8_t b = v & 0x7F;
    v >>= 7;
    if (v) {
      out.push_back((char)(b | 0x80));
    } else {
      out.push_back((char)b);
      return;
    }
  }
}

void put_tag(std::string& out, int fn, int wt) {
  put_varint(out, ((uint64_t)fn << 3) | wt);
}

void put_field_varint(std::string& out, int fn, int64_t v) {
  put_tag(out, fn, 0);
  put_varint(out, (uint64_t)v);
}

void put_field_float(std::string& out, int fn, float v) {
  put_tag(out, fn, 5);
  out.append(reinterpret_cast<const char*>(&v), 4);
}

void put_field_double(std::string& out, int fn, double v) {
  put_tag(out, fn, 1);
  out.append(reinterpret_cast<const char*>(&v), 8);
}

void put_field_bytes(std::string& out, int fn, const std::string& v) {
  put_tag(out, fn, 2);
  put_varint(out, v.size());
  out.append(v);
}

struct WireReader {
  const uint8_t* p;
  const uint8_t* end;
  explicit WireReader(const std::string& s)
      : p(reinterpret_cast<const uint8_t*>(s.data())),
        end(p + s.size()) {}
  bool done() const { return p >= end; }
  uint64_t varint() {
    uint64_t r = 0;
    int s = 0;
    while (p < end) {
      const uint8_t b = *p++;
      r |= (uint64_t)(b & 0x7F) << s;
      if (!(b & 0x80)) return r;
      s += 7;
    }
    throw std::runtime_error("truncated varint");
  }
  // Returns (field_number, wire_type); value read by caller.
  std::pair<int, int> tag() {
    const uint64_t t = varint();
    return {(int)(t >> 3), (int)(t & 7)};
  }
  float f32() {
    float v;
    std::memcpy(&v, p, 4);
    p += 4;
    return v;
  }
  double f64() {
    double v;
    std::memcpy(&v, p, 8);
    p += 8;
    return v;
  }
  std::string bytes() {
    const uint64_t n = varint();
    if (p + n > end) throw std::runtime_error("truncated bytes");
    std::string s(reinterpret_cast<const char*>(p), n);
    p += n;
    return s;
  }
  void skip(int wt) {
    if (wt == 0) {
      varint();
    } else if (wt == 1) {
      p += 8;
    } else if (wt == 2) {
      bytes();
    } else if (wt == 5) {
      p += 4;
    } else {
      throw std::runtime_error("bad wire type");
    }
  }
};

std::string read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) throw std::runtime_error("cannot open " + path);
  std::ostringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

void write_file(const std::string& path, const std::string& data) {
  std::ofstream f(path, std::ios::binary);
  if (!f) throw std::runtime_error("cannot write " + path);
  f.write(data.data(), (std::streamsize)data.size());
}

// ---------------------------------------------------------------------
// Data spec inference
// ---------------------------------------------------------------------
std::vector<float> quantile_boundaries(std::vector<float> v,
                                       int max_bins = 256) {
  v.erase(std::remove_if(v.begin(), v.end(),
                         [](float x) { return !std::isfinite(x); }),
          v.end());
  std::sort(v.begin(), v.end());
  v.erase(std::unique(v.begin(), v.end()), v.end());
  std::vector<float> cuts;
  if (v.size() <= 1) return cuts;
  if ((int)v.size() <= max_bins) {
    for (size_t i = 0; i + 1 < v.size(); ++i)
      cuts.push_back((float)(((double)v[i] + v[i + 1]) / 2.0));
  } else {
    for (int q = 1; q < max_bins; ++q) {
      const size_t idx = (size_t)((double)q / max_bins * (v.size() - 1));
      const float c = v[idx];
      if (cuts.empty() || c != cuts.back()) cuts.push_back(c);
    }
  }
  return cuts;
}

struct EncodedData {
  std::vector<ColumnSpec> specs;       // feature columns, sorted order
  std::vector<float> X;                // [F][N] feature-major
  std::vector<float> boundaries;       // [F][n_cuts] padded (+inf)
  int n_cuts = 1;
  std::vector<uint8_t> cat_flags;      // [F]
};

int vocab_index(const ColumnSpec& spec, const std::string& s) {
  for (size_t i = 0; i < spec.vocab.size(); ++i)
    if (spec.vocab[i] == s) return (int)i;
  return 0;
}

EncodedData encode_dataset(const Dataset& ds,
                           const std::vector<ColumnSpec>& specs) {
  EncodedData out;
  out.specs = specs;
  const size_t N = ds.num_rows();
  const size_t F = specs.size();
  out.X.assign(F * N, 0.f);
  out.cat_flags.assign(F, 0);
  for (size_t f = 0; f < F; ++f) {
    const auto& spec = specs[f];
    if (spec.type == ColumnType::kCategorical) {
      out.cat_flags[f] = 1;
      const auto it = ds.categoricals().find(spec.name);
      if (it == ds.categoricals().end())
        throw std::runtime_error("missing feature " + spec.name);
      std::map<std::string, int> lut;
      for (size_t i = 0; i < spec.vocab.size(); ++i)
        lut[spec.vocab[i]] = (int)i;
      for (size_t i = 0; i < N; ++i) {
        const auto vit = lut.find(it->second[i]);
        out.X[f * N + i] = (float)(vit == lut.end() ? 0 : vit->second);
      }
    } else {
      const auto it = ds.numericals().find(spec.name);
      if (it == ds.numericals().end())
        throw std::runtime_error("missing feature " + spec.name);
      for (size_t i = 0; i < N; ++i) {
        const float v = it->second[i];
        out.X[f * N + i] = std::isfinite(v) ? v : (float)spec.mean;
      }
    }
    out.n_cuts = std::max(out.n_cuts, (int)spec.boundaries.size());
  }
  out.n_cuts = std::min(out.n_cuts, 255);
  out.boundaries.assign(F * out.n_cuts,
                        std::numeric_limits<float>::infinity());
  for (size_t f = 0; f < F; ++f) {
    const auto& b = specs[f].boundaries;
    for (size_t j = 0; j < b.size() && (int)j < out.n_cuts; ++j)
      out.boundaries[f * out.n_cuts + j] = b[j];
  }
  return out;
}

std::vector<ColumnSpec> infer_specs(const Dataset& ds,
                                    const std::string& label) {
  std::vector<ColumnSpec> specs;
  for (const auto& [name, vals] : ds.numericals()) {
    if (name == label) continue;
    ColumnSpec s;
    s.name = name;
    s.type = ColumnType::kNumerical;
    double sum = 0.0;
    size_t n = 0;
    for (float v : vals)
      if (std::isfinite(v)) {
        sum += v;
        ++n;
      }
    s.mean = n ? sum / n : 0.0;
    s.boundaries = quantile_boundaries(vals);
    specs.push_back(std::move(s));
  }
  for (const auto& [name, vals] : ds.categoricals()) {
    if (name == label) continue;
    ColumnSpec s;
    s.name = name;
    s.type = ColumnType::kCategorical;
    std::map<std::string, int64_t> counts;
    for (const auto& v : vals) counts[v]++;
    std::vector<std::pair<int64_t, std::string>> order;
    for (const auto& [k, c] : counts) order.push_back({-c, k});
    std::sort(order.begin(), order.end());
    s.vocab.push_back("<OOD>");
    for (const auto& [negc, k] : order) {
      if (s.vocab.size() >= 256) break;  // u8 bin path
      s.vocab.push_back(k);
    }
    specs.push_back(std::move(s));
  }
  std::sort(specs.begin(), specs.end(),
            [](const ColumnSpec& a, const ColumnSpec& b) {
              return a.name < b.name;
            });
  return specs;
}

}  // namespace

// ---------------------------------------------------------------------
// Dataset
// ---------------------------------------------------------------------
void Dataset::AddNumerical(const std::string& name,
                           const std::vector<float>& values) {
  if (num_rows_ && values.size() != num_rows_)
    throw std::runtime_error("column size mismatch");
  num_rows_ = values.size();
  numericals_[name] = values;
}

void Dataset::AddCategorical(const std::string& name,
                             const std::vector<std::string>& values) {
  if (num_rows_ && values.size() != num_rows_)
    throw std::runtime_error("column size mismatch");
  num_rows_ = values.size();
  categoricals_[name] = values;
}

// ---------------------------------------------------------------------
// Training (level-wise dense loop over the cpu_* ops)
// ---------------------------------------------------------------------
std::unique_ptr<Model> TrainGradientBoostedTrees(
    const GbtConfig& cfg, const Dataset& ds, const std::string& label) {
  const int64_t N = (int64_t)ds.num_rows();
  if (N == 0) throw std::runtime_error("empty dataset");

  auto model = std::make_unique<Model>();
  model->label_ = label;

  // label
  std::vector<float> y(N);
  const auto nit = ds.numericals().find(label);
  if (nit != ds.numericals().end()) {
    model->task_ = Task::kRegression;
    y.assign(nit->second.begin(), nit->second.end());
  } else {
    const auto cit = ds.categoricals().find(label);
    if (cit == ds.categoricals().end())
      throw std::runtime_error("label column not found: " + label);
    std::map<std::string, int64_t> counts;
    for (const auto& v : cit->second) counts[v]++;
    if (counts.size() != 2)
      throw std::runtime_error(
          "C++ API supports binary classification / regression");
    // classes ordered by ascending frequency? Reference vocab is
    // most-frequent-first; positive class = vocab index 2's item.
    std::vector<std::pair<int64_t, std::string>> order;
    for (const auto& [k, c] : counts) order.push_back({-c, k});
    std::sort(order.begin(), order.end());
    model->label_classes_ = {order[0].second, order[1].second};
    model->task_ = Task::kClassification;
    model->sigmoid_ = true;
    for (int64_t i = 0; i < N; ++i)
      y[i] = cit->second[i] == order[1].second ? 1.f : 0.f;
  }

  auto specs = infer_specs(ds, label);
  auto enc = encode_dataset(ds, specs);
  const int F = (int)specs.size();
  model->features_ = specs;

  // bin
  std::vector<uint8_t> bins(F * N);
  cpu_bin_data(enc.X.data(), enc.boundaries.data(), bins.data(), N, F,
               enc.n_cuts, /*na_to_255=*/0);
  // categorical columns: the code IS the bin
  for (int f = 0; f < F; ++f)
    if (enc.cat_flags[f])
      for (int64_t i = 0; i < N; ++i) {
        float v = enc.X[(int64_t)f * N + i];
        bins[(int64_t)f * N + i] =
            (uint8_t)std::min(std::max(0, (int)v), 255);
      }

  const int n_bins = 256;
  const int max_depth = cfg.max_depth;
  const int total_nodes = (1 << (max_depth + 1)) - 1;
  const int max_level = 1 << (max_depth - 1 + 1);  // widest split level

  // initial prediction
  float init;
  if (model->task_ == Task::kClassification) {
    double p = 0;
    for (int64_t i = 0; i < N; ++i) p += y[i];
    p = std::min(std::max(p / N, 1e-6), 1.0 - 1e-6);
    init = (float)std::log(p / (1.0 - p));
  } else {
    double m = 0;
    for (int64_t i = 0; i < N; ++i) m += y[i];
    init = (float)(m / N);
  }
  model->init_prediction_ = init;
  model->leaf_scale_ = 1.0f;  // shrinkage baked into leaf values

  std::vector<float> preds(N, init);
  std::vector<float> gh(2 * N);
  std::vector<int32_t> node_ids(N);
  std::vector<float> hist((size_t)max_level * F * n_bins * 3);
  std::vector<int32_t> slot_map(max_level);
  std::vector<int32_t> abs_of_slot(max_level);
  std::vector<float> node_stats(total_nodes * 3);
  std::vector<float> bg_nf((size_t)max_level * F);
  std::vector<int32_t> bb_nf((size_t)max_level * F);
  std::vector<int32_t> best_feat(max_level), best_bin(max_level);
  std::vector<float> best_gain(max_level);
  std::vector<float> leaf_vals(total_nodes);
  std::vector<unsigned long long> tree_masks(
      (size_t)total_nodes * (kMaxBins / 64));
  std::vector<int32_t> tree_feat(total_nodes), tree_bin(total_nodes);

  SplitParams sp;
  sp.lambda_l2 = cfg.l2_regularization;
  sp.lambda_l1 = cfg.l1_regularization;
  sp.min_hessian = cfg.min_sum_hessian_in_leaf;
  sp.min_examples = cfg.min_examples;
  sp.min_gain = 0.f;
  sp.cat_smooth = cfg.l2_categorical_regularization;
  sp.na_mode = 0;

  const int loss = model->task_ == Task::kClassification
                       ? kLossBinomial
                       : kLossSquaredError;
  const bool any_cat =
      std::any_of(enc.cat_flags.begin(), enc.cat_flags.end(),
                  [](uint8_t v) { return v != 0; });

  for (int t = 0; t < cfg.num_trees; ++t) {
    cpu_grad_hess(preds.data(), y.data(), gh.data(), N, loss);
    std::fill(node_ids.begin(), node_ids.end(), 0);
    std::fill(node_stats.begin(), node_stats.end(), 0.f);
    std::fill(tree_masks.begin(), tree_masks.end(), 0ull);
    std::fill(tree_feat.begin(), tree_feat.end(), -1);
    for (int level = 0; level < max_depth; ++level) {
      const int level_base = (1 << level) - 1;
      const int level_size = 1 << level;
      std::fill_n(hist.begin(), (size_t)level_size * F * n_bins * 3,
                  0.f);
      for (int r = 0; r < level_size; ++r) {
        slot_map[r] = r;
        abs_of_slot[r] = level_base + r;
      }
      cpu_hist_build(bins.data(), gh.data(), node_ids.data(),
                     slot_map.data(), hist.data(), N, F, n_bins,
                     level_base, level_size, 0, level_size);
      cpu_split_scan(hist.data(), abs_of_slot.data(), node_stats.data(),
                     bg_nf.data(), bb_nf.data(), best_feat.data(),
                     best_bin.data(), best_gain.data(),
                     /*feat_mask=*/nullptr,
                     any_cat ? enc.cat_flags.data() : nullptr,
                     tree_masks.data(), /*mono=*/nullptr,
                     /*node_bounds=*/nullptr, /*na_meanb=*/nullptr,
                     /*tree_na=*/nullptr, F, n_bins, 0, level_size, sp);
      for (int r = 0; r < level_size; ++r) {
        tree_feat[level_base + r] = best_feat[r];
        tree_bin[level_base + r] = best_bin[r];
      }
      cpu_update_node_ids(bins.data(), node_ids.data(), slot_map.data(),
                          best_feat.data(), best_bin.data(),
                          any_cat ? enc.cat_flags.data() : nullptr,
                          tree_masks.data(), /*tree_na=*/nullptr, N,
                          level_base, level_size);
    }
    cpu_leaf_values(node_stats.data(), /*bounds=*/nullptr,
                    leaf_vals.data(), total_nodes, cfg.l2_regularization,
                    cfg.l1_regularization);
    cpu_update_preds(preds.data(), node_ids.data(), leaf_vals.data(), N,
                     cfg.shrinkage);

    // flatten the implicit tree into the model (pre-order-compatible
    // SoA layout; right = left + 1), leaves scaled by shrinkage
    const int root = (int)model->feat_.size();
    model->roots_.push_back(root);
    // BFS over reachable nodes (ascending index == level order)
    std::vector<int> nodes;
    std::vector<char> reach(total_nodes, 0);
    reach[0] = 1;
    for (int a = 0; a < total_nodes; ++a) {
      if (!reach[a]) continue;
      nodes.push_back(a);
      if (tree_feat[a] >= 0 && 2 * a + 2 < total_nodes) {
        reach[2 * a + 1] = 1;
        reach[2 * a + 2] = 1;
      }
    }
    std::vector<int32_t> new_idx(total_nodes, -1);
    for (size_t k = 0; k < nodes.size(); ++k)
      new_idx[nodes[k]] = (int32_t)k;
    for (int a : nodes) {
      const bool internal = tree_feat[a] >= 0 && 2 * a + 2 < total_nodes
                            && new_idx[2 * a + 1] >= 0;
      if (internal) {
        const int f = tree_feat[a];
        model->feat_.push_back(f);
        model->left_.push_back(root + new_idx[2 * a + 1]);
        if (enc.cat_flags[f]) {
          model->cat_idx_.push_back((int32_t)(model->masks_.size() / 4));
          for (int w = 0; w < 4; ++w)
            model->masks_.push_back(
                tree_masks[(size_t)a * 4 + w]);
          model->thr_.push_back(0.f);
        } else {
          model->cat_idx_.push_back(-1);
          model->thr_.push_back(
              enc.boundaries[(size_t)f * enc.n_cuts + tree_bin[a]]);
        }
      } else {
        model->feat_.push_back(-1);
        model->left_.push_back(0);
        model->cat_idx_.push_back(-1);
        model->thr_.push_back(cfg.shrinkage * leaf_vals[a]);
      }
      model->cover_.push_back(node_stats[(size_t)a * 3 + 2]);
    }
  }
  return model;
}

// ---------------------------------------------------------------------
// Prediction
// ---------------------------------------------------------------------
float Model::PredictRow(const float* f) const {
  float acc = init_prediction_;
  for (size_t t = 0; t < roots_.size(); ++t) {
    int n = roots_[t];
    while (feat_[n] >= 0) {
      const float x = f[feat_[n]];
      bool right;
      const int32_t ci = cat_idx_[n];
      if (ci >= 0) {
        const int v = x < 0 ? 0 : (x > 255 ? 255 : (int)x);
        right = (masks_[(size_t)ci * 4 + (v >> 6)] >> (v & 63)) & 1ull;
      } else {
        right = x > thr_[n];
      }
      n = left_[n] + (right ? 1 : 0);
    }
    float leaf = thr_[n];
    if (is_rf_ && winner_take_all_) leaf = leaf > 0.5f ? 1.f : 0.f;
    acc += leaf * leaf_scale_;
  }
  if (sigmoid_) acc = 1.0f / (1.0f + std::exp(-acc));
  return acc;
}

std::vector<float> Model::Predict(const Dataset& ds) const {
  const size_t N = ds.num_rows();
  const size_t F = features_.size();
  // resolve column pointers + vocab lookup tables ONCE (not per row)
  struct Col {
    const std::vector<float>* num = nullptr;
    const std::vector<std::string>* cat = nullptr;
    std::map<std::string, int> lut;
    float mean = 0.f;
  };
  std::vector<Col> cols(F);
  for (size_t f = 0; f < F; ++f) {
    const auto& spec = features_[f];
    if (spec.type == ColumnType::kCategorical) {
      const auto it = ds.categoricals().find(spec.name);
      if (it == ds.categoricals().end())
        throw std::runtime_error("missing feature " + spec.name);
      cols[f].cat = &it->second;
      for (size_t i = 0; i < spec.vocab.size(); ++i)
        cols[f].lut[spec.vocab[i]] = (int)i;
    } else {
      const auto it = ds.numericals().find(spec.name);
      if (it == ds.numericals().end())
        throw std::runtime_error("missing feature " + spec.name);
      cols[f].num = &it->second;
      cols[f].mean = (float)spec.mean;
    }
  }
  std::vector<float> row(F);
  std::vector<float> out(N);
  for (size_t i = 0; i < N; ++i) {
    for (size_t f = 0; f < F; ++f) {
      if (cols[f].cat != nullptr) {
        const auto vit = cols[f].lut.find((*cols[f].cat)[i]);
        row[f] = (float)(vit == cols[f].lut.end() ? 0 : vit->second);
      } else {
        const float v = (*cols[f].num)[i];
        row[f] = std::isfinite(v) ? v : cols[f].mean;
      }
    }
    out[i] = PredictRow(row.data());
  }
  return out;
}

// ---------------------------------------------------------------------
// Save (reference model directory)
// ---------------------------------------------------------------------
namespace {

std::string encode_data_spec(const Model& m) {
  std::string cols;
  auto add_col = [&](const std::string& body) {
    std::string out;
    put_field_bytes(out, 1, body);
    cols += out;
  };
  for (const auto& spec : m.features_) {
    std::string body;
    if (spec.type == ColumnType::kCategorical) {
      put_field_varint(body, 1, 4);  // ColumnType CATEGORICAL
      put_field_bytes(body, 2, spec.name);
      std::string cat;
      put_field_varint(cat, 2, (int64_t)spec.vocab.size());
      for (size_t i = 0; i < spec.vocab.size(); ++i) {
        std::string vv;
        put_field_varint(vv, 1, (int64_t)i);
        std::string entry;
        put_field_bytes(entry, 1, spec.vocab[i]);
        put_field_bytes(entry, 2, vv);
        put_field_bytes(cat, 7, entry);  // items map entry
      }
      put_field_bytes(body, 6, cat);
    } else {
      put_field_varint(body, 1, 1);  // NUMERICAL
      put_field_bytes(body, 2, spec.name);
      std::string num;
      put_field_double(num, 1, spec.mean);
      put_field_bytes(body, 5, num);
    }
    add_col(body);
  }
  // label column
  std::string body;
  if (m.task_ == Task::kClassification) {
    put_field_varint(body, 1, 4);
    put_field_bytes(body, 2, m.label_);
    std::string cat;
    put_field_varint(cat, 2, (int64_t)m.label_classes_.size() + 1);
    std::vector<std::string> vocab = {"<OOD>"};
    vocab.insert(vocab.end(), m.label_classes_.begin(),
                 m.label_classes_.end());
    for (size_t i = 0; i < vocab.size(); ++i) {
      std::string vv;
      put_field_varint(vv, 1, (int64_t)i);
      std::string entry;
      put_field_bytes(entry, 1, vocab[i]);
      put_field_bytes(entry, 2, vv);
      put_field_bytes(cat, 7, entry);
    }
    put_field_bytes(body, 6, cat);
  } else {
    put_field_varint(body, 1, 1);
    put_field_bytes(body, 2, m.label_);
    std::string num;
    put_field_double(num, 1, 0.0);
    put_field_bytes(body, 5, num);
  }
  add_col(body);
  return cols;
}

void encode_node(const Model& m, int n, std::string& records_out,
                 std::vector<std::string>& records) {
  std::string body;
  if (m.feat_[n] < 0) {
    std::string reg;
    put_field_float(reg, 1, m.thr_[n]);
    put_field_bytes(body, 2, reg);  // Node.regressor.top_value
  } else {
    std::string reg;
    put_field_float(reg, 1, 0.f);
    put_field_bytes(body, 2, reg);
    std::string inner;
    if (m.cat_idx_[n] >= 0) {
      std::string bm(reinterpret_cast<const char*>(
                         &m.masks_[(size_t)m.cat_idx_[n] * 4]),
                     32);
      std::string cb;
      put_field_bytes(cb, 1, bm);
      put_field_bytes(inner, 5, cb);  // ContainsBitmap
    } else {
      // our split is x > thr; reference Higher is x >= thr:
      // nextafter up
      float t = std::nextafter(m.thr_[n],
                               std::numeric_limits<float>::infinity());
      std::string hi;
      put_field_float(hi, 1, t);
      put_field_bytes(inner, 2, hi);  // Higher
    }
    std::string cond;
    put_field_varint(cond, 2, m.feat_[n]);  // attribute
    put_field_bytes(cond, 3, inner);
    put_field_varint(cond, 4,
                     (int64_t)std::max(0.f, m.cover_[n]));
    put_field_double(cond, 5, m.cover_[n]);
    put_field_bytes(body, 3, cond);  // Node.condition
  }
  records.push_back(body);
  if (m.feat_[n] >= 0) {
    encode_node(m, m.left_[n], records_out, records);      // negative
    encode_node(m, m.left_[n] + 1, records_out, records);  // positive
  }
}

}  // namespace

void Model::Save(const std::string& dir) const {
  if (is_rf_)
    throw std::runtime_error("C++ Save supports GBT models");
  // header.pb: name=1, task=2, label_col_idx=3, input_features=5
  std::string header;
  put_field_bytes(header, 1, "GRADIENT_BOOSTED_TREES");
  put_field_varint(header, 2,
                   task_ == Task::kClassification ? 1 : 2);
  put_field_varint(header, 3, (int64_t)features_.size());
  for (size_t i = 0; i < features_.size(); ++i)
    put_field_varint(header, 5, (int64_t)i);
  write_file(dir + "/header.pb", header);
  write_file(dir + "/data_spec.pb", encode_data_spec(*this));
  // gradient_boosted_trees_header.pb
  std::string gh;
  put_field_varint(gh, 1, 1);  // num_node_shards
  put_field_varint(gh, 2, (int64_t)roots_.size());
  put_field_varint(gh, 3, task_ == Task::kClassification ? 1 : 2);
  put_field_float(gh, 4, init_prediction_);
  put_field_varint(gh, 5, 1);  // num_trees_per_iter
  put_field_bytes(gh, 7, "BLOB_SEQUENCE");
  write_file(dir + "/gradient_boosted_trees_header.pb", gh);
  // nodes blob sequence
  std::vector<std::string> records;
  std::string dummy;
  for (size_t t = 0; t < roots_.size(); ++t)
    encode_node(*this, roots_[t], dummy, records);
  std::string blob("BS", 2);
  const char hdr[6] = {1, 0, 0, 0, 0, 0};
  blob.append(hdr, 6);
  for (const auto& r : records) {
    const uint32_t len = (uint32_t)r.size();
    blob.append(reinterpret_cast<const char*>(&len), 4);
    blob.append(r);
  }
  write_file(dir + "/nodes-00000-of-00001", blob);
  write_file(dir + "/done", "");
}

// ---------------------------------------------------------------------
// Load (reference model directory)
// ---------------------------------------------------------------------
namespace {

struct ParsedNode {
  bool leaf = true;
  float value = 0.f;     // leaf value / threshold
  int attr = -1;
  bool is_cat = false;
  uint64_t mask[4] = {0, 0, 0, 0};
  double cover = 0.0;
  float cls_p = 0.f;     // RF classification leaf probability
  bool has_cls = false;
};

ParsedNode parse_node(const std::string& rec) {
  ParsedNode out;
  WireReader r(rec);
  while (!r.done()) {
    auto [fn, wt] = r.tag();
    if (fn == 2 && wt == 2) {  // regressor
      const std::string sub = r.bytes();
      WireReader s(sub);
      while (!s.done()) {
        auto [f2, w2] = s.tag();
        if (f2 == 1 && w2 == 5)
          out.value = s.f32();
        else
          s.skip(w2);
      }
    } else if (fn == 1 && wt == 2) {  // classifier
      const std::string sub = r.bytes();
      WireReader s(sub);
      while (!s.done()) {
        auto [f2, w2] = s.tag();
        if (f2 == 2 && w2 == 2) {  // distribution
          const std::string d = s.bytes();
          WireReader ds_(d);
          std::vector<double> counts;
          double total = 0;
          while (!ds_.done()) {
            auto [f3, w3] = ds_.tag();
            if (f3 == 1 && w3 == 2) {
              const std::string packed = ds_.bytes();
              for (size_t i = 0; i + 8 <= packed.size(); i += 8) {
                double v;
                std::memcpy(&v, packed.data() + i, 8);
                counts.push_back(v);
              }
            } else if (f3 == 2 && w3 == 1) {
              total = ds_.f64();
            } else {
              ds_.skip(w3);
            }
          }
          if (total > 0 && counts.size() >= 3) {
            out.cls_p = (float)(counts[2] / total);
            out.has_cls = true;
          }
        } else {
          s.skip(w2);
        }
      }
    } else if (fn == 3 && wt == 2) {  // condition
      out.leaf = false;
      const std::string cond = r.bytes();
      WireReader c(cond);
      while (!c.done()) {
        auto [f2, w2] = c.tag();
        if (f2 == 2 && w2 == 0) {
          out.attr = (int)c.varint();
        } else if (f2 == 5 && w2 == 1) {
          out.cover = c.f64();
        } else if (f2 == 3 && w2 == 2) {
          const std::string inner = c.bytes();
          WireReader ic(inner);
          while (!ic.done()) {
            auto [f3, w3] = ic.tag();
            if (f3 == 2 && w3 == 2) {  // Higher
              const std::string hi = ic.bytes();
              WireReader h(hi);
              while (!h.done()) {
                auto [f4, w4] = h.tag();
                if (f4 == 1 && w4 == 5)
                  // reference >= t  <=>  our > nextafter(t, -inf)
                  out.value = std::nextafter(
                      h.f32(),
                      -std::numeric_limits<float>::infinity());
                else
                  h.skip(w4);
              }
            } else if (f3 == 5 && w3 == 2) {  // ContainsBitmap
              out.is_cat = true;
              const std::string cb = ic.bytes();
              WireReader b(cb);
              while (!b.done()) {
                auto [f4, w4] = b.tag();
                if (f4 == 1 && w4 == 2) {
                  const std::string bm = b.bytes();
                  std::memcpy(out.mask, bm.data(),
                              std::min<size_t>(bm.size(), 32));
                } else {
                  b.skip(w4);
                }
              }
            } else if (f3 == 4 && w3 == 2) {  // ContainsVector
              out.is_cat = true;
              const std::string cv = ic.bytes();
              WireReader b(cv);
              while (!b.done()) {
                auto [f4, w4] = b.tag();
                if (f4 == 1 && w4 == 2) {
                  const std::string packed = b.bytes();
                  WireReader e(packed);
                  while (!e.done()) {
                    const uint64_t el = e.varint();
                    if (el < 256)
                      out.mask[el >> 6] |= 1ull << (el & 63);
                  }
                } else if (f4 == 1 && w4 == 0) {
                  const uint64_t el = b.varint();
                  if (el < 256) out.mask[el >> 6] |= 1ull << (el & 63);
                } else {
                  b.skip(w4);
                }
              }
            } else {
              throw std::runtime_error(
                  "C++ Load: unsupported condition type (field " +
                  std::to_string(f3) + ")");
            }
          }
        } else {
          c.skip(w2);
        }
      }
    } else {
      r.skip(wt);
    }
  }
  return out;
}

}  // namespace

std::unique_ptr<Model> Model::Load(const std::string& dir) {
  auto m = std::make_unique<Model>();
  // header.pb
  {
    std::string hdr_str = read_file(dir + "/header.pb");
    WireReader h(hdr_str);
    std::string name;
    int64_t task = 1, label_idx = -1;
    std::vector<int> input_features;
    while (!h.done()) {
      auto [fn, wt] = h.tag();
      if (fn == 1 && wt == 2)
        name = h.bytes();
      else if (fn == 2 && wt == 0)
        task = (int64_t)h.varint();
      else if (fn == 3 && wt == 0)
        label_idx = (int64_t)h.varint();
      else if (fn == 5 && wt == 0)
        input_features.push_back((int)h.varint());
      else
        h.skip(wt);
    }
    m->is_rf_ = name == "RANDOM_FOREST";
    m->task_ = task == 2 ? Task::kRegression : Task::kClassification;
    m->sigmoid_ = false;

    // data_spec.pb
    std::vector<ColumnSpec> all_cols;
    std::string ds_str = read_file(dir + "/data_spec.pb");
    WireReader dd(ds_str);
    while (!dd.done()) {
      auto [fn, wt] = dd.tag();
      if (fn == 1 && wt == 2) {
        const std::string col = dd.bytes();
        WireReader c(col);
        ColumnSpec spec;
        while (!c.done()) {
          auto [f2, w2] = c.tag();
          if (f2 == 1 && w2 == 0) {
            const uint64_t t = c.varint();
            spec.type = (t == 4 || t == 5)
                            ? ColumnType::kCategorical
                            : (t == 7 ? ColumnType::kBoolean
                                      : ColumnType::kNumerical);
          } else if (f2 == 2 && w2 == 2) {
            spec.name = c.bytes();
          } else if (f2 == 5 && w2 == 2) {
            const std::string num = c.bytes();
            WireReader nn(num);
            while (!nn.done()) {
              auto [f3, w3] = nn.tag();
              if (f3 == 1 && w3 == 1)
                spec.mean = nn.f64();
              else if (f3 == 1 && w3 == 5)
                spec.mean = nn.f32();
              else
                nn.skip(w3);
            }
          } else if (f2 == 6 && w2 == 2) {
            const std::string cat = c.bytes();
            WireReader cc(cat);
            std::map<int64_t, std::string> by_idx;
            while (!cc.done()) {
              auto [f3, w3] = cc.tag();
              if (f3 == 7 && w3 == 2) {
                const std::string entry = cc.bytes();
                WireReader e(entry);
                std::string key;
                int64_t idx = 0;
                while (!e.done()) {
                  auto [f4, w4] = e.tag();
                  if (f4 == 1 && w4 == 2) {
                    key = e.bytes();
                  } else if (f4 == 2 && w4 == 2) {
                    const std::string vv = e.bytes();
                    WireReader v(vv);
                    while (!v.done()) {
                      auto [f5, w5] = v.tag();
                      if (f5 == 1 && w5 == 0)
                        idx = (int64_t)v.varint();
                      else
                        v.skip(w5);
                    }
                  } else {
                    e.skip(w4);
                  }
                }
                by_idx[idx] = key;
              } else {
                cc.skip(w3);
              }
            }
            if (!by_idx.empty()) {
              spec.vocab.resize(by_idx.rbegin()->first + 1);
              for (const auto& [i, k] : by_idx) spec.vocab[i] = k;
            }
          } else {
            c.skip(w2);
          }
        }
        all_cols.push_back(std::move(spec));
      } else {
        dd.skip(wt);
      }
    }
    for (int ci : input_features)
      if (ci >= 0 && ci < (int)all_cols.size())
        m->features_.push_back(all_cols[ci]);
    if (label_idx >= 0 && label_idx < (int64_t)all_cols.size()) {
      m->label_ = all_cols[label_idx].name;
      const auto& lv = all_cols[label_idx].vocab;
      if (lv.size() >= 3)
        m->label_classes_.assign(lv.begin() + 1, lv.end());
    }
    // remap: original column index -> dense feature index
    std::map<int, int> remap;
    for (size_t i = 0; i < input_features.size(); ++i)
      remap[input_features[i]] = (int)i;

    // family header
    float init = 0.f;
    std::ifstream gf(dir + "/gradient_boosted_trees_header.pb",
                     std::ios::binary);
    if (gf) {
      std::ostringstream ss;
      ss << gf.rdbuf();
      std::string gh_str = ss.str();
      WireReader g(gh_str);
      while (!g.done()) {
        auto [fn2, wt2] = g.tag();
        if (fn2 == 4 && wt2 == 5)
          init = g.f32();
        else if (fn2 == 3 && wt2 == 0) {
          const uint64_t lv = g.varint();
          m->sigmoid_ = lv == 1;  // BINOMIAL_LOG_LIKELIHOOD
        } else
          g.skip(wt2);
      }
      m->leaf_scale_ = 1.0f;
    } else if (m->is_rf_) {
      std::string rh_str = read_file(dir + "/random_forest_header.pb");
      WireReader g(rh_str);
      while (!g.done()) {
        auto [fn2, wt2] = g.tag();
        if (fn2 == 3 && wt2 == 0)
          m->winner_take_all_ = g.varint() != 0;
        else
          g.skip(wt2);
      }
    } else {
      throw std::runtime_error("unsupported model family in " + dir);
    }
    m->init_prediction_ = init;

    // nodes
    std::string blob = read_file(dir + "/nodes-00000-of-00001");
    if (blob.size() < 8 || blob[0] != 'B' || blob[1] != 'S')
      throw std::runtime_error("bad blob-sequence magic");
    std::vector<std::string> records;
    size_t pos = 8;
    while (pos + 4 <= blob.size()) {
      uint32_t len;
      std::memcpy(&len, blob.data() + pos, 4);
      pos += 4;
      records.push_back(blob.substr(pos, len));
      pos += len;
    }
    // rebuild trees: records are pre-order (negative child first);
    // two-phase — parse into a temp tree, then lay out BFS so the flat
    // arrays keep the right == left + 1 invariant
    size_t rec_pos = 0;
    struct TmpNode {
      ParsedNode pn;
      int neg = -1, pos_ = -1;
    };
    std::vector<TmpNode> tmp;
    std::function<int(void)> parse_rec = [&]() -> int {
      const int my = (int)tmp.size();
      tmp.push_back({parse_node(records.at(rec_pos++)), -1, -1});
      if (!tmp[my].pn.leaf) {
        const int a = parse_rec();
        const int b = parse_rec();
        tmp[my].neg = a;
        tmp[my].pos_ = b;
      }
      return my;
    };
    // count trees from the family header is not strictly needed: keep
    // parsing trees until all records are consumed
    std::vector<int> tmp_roots;
    while (rec_pos < records.size()) tmp_roots.push_back(parse_rec());
    // layout: BFS per tree, children adjacent
    for (int tr : tmp_roots) {
      const int root = (int)m->feat_.size();
      m->roots_.push_back(root);
      std::vector<int> queue = {tr};
      std::vector<int> placed(tmp.size(), -1);
      // first pass: assign indices
      for (size_t qi = 0; qi < queue.size(); ++qi) {
        const int a = queue[qi];
        placed[a] = root + (int)qi;
        if (tmp[a].neg >= 0) {
          queue.push_back(tmp[a].neg);
          queue.push_back(tmp[a].pos_);
        }
      }
      for (size_t qi = 0; qi < queue.size(); ++qi) {
        const int a = queue[qi];
        const ParsedNode& pn = tmp[a].pn;
        if (pn.leaf) {
          m->feat_.push_back(-1);
          m->thr_.push_back(pn.has_cls ? pn.cls_p : pn.value);
          m->left_.push_back(0);
          m->cat_idx_.push_back(-1);
        } else {
          const auto rit = remap.find(pn.attr);
          m->feat_.push_back(rit == remap.end() ? 0 : rit->second);
          m->left_.push_back(placed[tmp[a].neg]);
          if (pn.is_cat) {
            m->cat_idx_.push_back((int32_t)(m->masks_.size() / 4));
            for (int w = 0; w < 4; ++w) m->masks_.push_back(pn.mask[w]);
            m->thr_.push_back(0.f);
          } else {
            m->cat_idx_.push_back(-1);
            m->thr_.push_back(pn.value);
          }
        }
        m->cover_.push_back((float)pn.cover);
      }
    }
    if (m->is_rf_)
      m->leaf_scale_ = m->roots_.empty()
                           ? 1.f
                           : 1.0f / (float)m->roots_.size();
  }
  return m;
}

}  // namespace api
}  // namespace ydfa
