// CPU implementations of the training/inference ops. Semantics match the
// HIP kernels bit-for-bit where possible (identical tie-break rules,
// identical gain formula); floating-point reduction order differs (sequential
// here vs. LDS trees on GPU), which the tests cover with tolerances.
// Capability analogue of the reference's threaded splitter
// (learner/decision_tree/training.cc StreamProcessor jobs); parallelised
// over features / row blocks with a persistent thread pool.
#include <algorithm>
#include <atomic>
#include <cmath>
#include <condition_variable>
#include <cstdint>
#include <cstring>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

#include "common.h"

namespace ydfa {

// ---------------------------------------------------------------------------
// Minimal persistent thread pool (reference analogue: utils/concurrency.h
// ThreadPool). Static singleton, sized once from hardware_concurrency.
// ---------------------------------------------------------------------------
class ThreadPool {
 public:
  static ThreadPool& Get() {
    static ThreadPool pool;
    return pool;
  }

  int size() const { return (int)workers_.size(); }

  // Runs fn(block_idx) for block_idx in [0, n_blocks), blocking until done.
  void ParallelFor(int n_blocks, const std::function<void(int)>& fn) {
    if (n_blocks <= 1 || workers_.empty()) {
      for (int i = 0; i < n_blocks; ++i) fn(i);
      return;
    }
    std::unique_lock<std::mutex> lk(m_);
    fn_ = &fn;
    next_.store(0);
    remaining_.store(n_blocks);
    n_blocks_ = n_blocks;
    ++epoch_;
    cv_.notify_all();
    // Wait until all blocks ran AND every worker left the work loop — a
    // straggler inside the loop must not observe the next epoch's counters.
    done_cv_.wait(lk,
                  [&] { return remaining_.load() == 0 && active_ == 0; });
    fn_ = nullptr;
  }

 private:
  ThreadPool() {
    int n = (int)std::thread::hardware_concurrency();
    if (n < 1) n = 1;
    if (n > 64) n = 64;
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this] { WorkerLoop(); });
  }
  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> lk(m_);
      stop_ = true;
      cv_.notify_all();
    }
    for (auto& t : workers_) t.join();
  }

  void WorkerLoop() {
    uint64_t seen = 0;
    for (;;) {
      const std::function<void(int)>* fn;
      {
        std::unique_lock<std::mutex> lk(m_);
        cv_.wait(lk, [&] { return stop_ || (fn_ && epoch_ != seen); });
        if (stop_) return;
        seen = epoch_;
        fn = fn_;
        ++active_;
      }
      for (;;) {
        int i = next_.fetch_add(1);
        if (i >= n_blocks_) break;
        (*fn)(i);
        remaining_.fetch_sub(1);
      }
      {
        std::lock_guard<std::mutex> lk(m_);
        --active_;
        done_cv_.notify_all();
      }
    }
  }

  std::vector<std::thread> workers_;
  std::mutex m_;
  std::condition_variable cv_, done_cv_;
  const std::function<void(int)>* fn_ = nullptr;
  std::atomic<int> next_{0}, remaining_{0};
  int active_ = 0;
  int n_blocks_ = 0;
  uint64_t epoch_ = 0;
  bool stop_ = false;
};

extern "C" {

void cpu_bin_data(const float* x, const float* boundaries, uint8_t* out,
                  int64_t N, int F, int n_cuts, int na_to_255) {
  ThreadPool::Get().ParallelFor(F, [&](int f) {
    const float* bnd = boundaries + (int64_t)f * n_cuts;
    const float* xf = x + (int64_t)f * N;
    uint8_t* of = out + (int64_t)f * N;
    for (int64_t i = 0; i < N; ++i) {
      const float v = xf[i];
      if (na_to_255 && std::isnan(v)) {
        of[i] = 255;
        continue;
      }
      int lo = 0, hi = n_cuts;
      while (lo < hi) {
        const int mid = (lo + hi) >> 1;
        if (bnd[mid] < v) lo = mid + 1; else hi = mid;
      }
      of[i] = (uint8_t)lo;
    }
  });
}

void cpu_grad_hess(const float* preds, const float* labels, float* gh,
                   int64_t N, int loss) {
  const int nb = std::max(1, std::min<int>(ThreadPool::Get().size(),
                                           (int)(N / 16384) + 1));
  const int64_t per = (N + nb - 1) / nb;
  ThreadPool::Get().ParallelFor(nb, [&](int blk) {
    const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
    for (int64_t k = i0; k < i1; ++k) {
      float g, h;
      if (loss == kLossBinomial) {
        const float p = 1.0f / (1.0f + std::exp(-preds[k]));
        g = p - labels[k];
        h = std::max(p * (1.0f - p), 1e-16f);
      } else if (loss == kLossPoisson) {
        const float ep = std::exp(std::min(preds[k], 15.0f));
        g = ep - labels[k];
        h = std::max(ep, 1e-6f);
      } else if (loss == kLossMAE) {
        g = (preds[k] > labels[k]) ? 1.0f : -1.0f;
        h = 1.0f;
      } else {
        g = preds[k] - labels[k];
        h = 1.0f;
      }
      gh[2 * k] = g;
      gh[2 * k + 1] = h;
    }
  });
}

void cpu_grad_hess_softmax(const float* preds, const float* labels, float* gh,
                           int64_t N, int n_classes, int cls) {
  const int nb = std::max(1, std::min<int>(ThreadPool::Get().size(),
                                           (int)(N / 8192) + 1));
  const int64_t per = (N + nb - 1) / nb;
  ThreadPool::Get().ParallelFor(nb, [&](int blk) {
    const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
    for (int64_t k = i0; k < i1; ++k) {
      float m = -1e30f;
      for (int c = 0; c < n_classes; ++c)
        m = std::max(m, preds[(int64_t)c * N + k]);
      float denom = 0.f;
      for (int c = 0; c < n_classes; ++c)
        denom += std::exp(preds[(int64_t)c * N + k] - m);
      const float p = std::exp(preds[(int64_t)cls * N + k] - m) / denom;
      const float y = (labels[k] == (float)cls) ? 1.0f : 0.0f;
      gh[2 * k] = p - y;
      gh[2 * k + 1] = std::max(p * (1.0f - p), 1e-16f);
    }
  });
}

void cpu_weighted_target(const float* labels, const float* weights, float* gh,
                         int64_t N) {
  for (int64_t k = 0; k < N; ++k) {
    const float w = weights ? weights[k] : 1.0f;
    gh[2 * k] = -w * labels[k];
    gh[2 * k + 1] = w;
  }
}

// hist: [n_slots][F][n_bins][3], pre-zeroed by caller; hist base pointer
// corresponds to slot0. Count accumulates examples with h != 0 (zero-weight
// bootstrap rows are out-of-bag).
void cpu_hist_build(const uint8_t* bins, const float* gh,
                    const int32_t* node_ids, const int32_t* slot_map,
                    float* hist, int64_t N, int F, int n_bins, int level_base,
                    int level_size, int slot0, int n_slots) {
  ThreadPool::Get().ParallelFor(F, [&](int f) {
    const uint8_t* fb = bins + (int64_t)f * N;
    for (int64_t i = 0; i < N; ++i) {
      const int rel = node_ids[i] - level_base;
      if (rel < 0 || rel >= level_size) continue;
      const int slot = slot_map[rel] - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      float* p = hist + (((int64_t)slot * F + f) * n_bins + fb[i]) * 3;
      const float h = gh[2 * i + 1];
      p[0] += gh[2 * i];
      p[1] += h;
      if (h != 0.f) p[2] += 1.0f;
    }
  });
}

// Categories ordered ascending by (G/(H+smooth), bin); empty bins last.
// Matches bitonic_sort_bins in train_kernels.hip.
static void sorted_bin_order(const float* hp, int n_bins, float smooth,
                             std::vector<int>& order) {
  order.resize(n_bins);
  std::vector<float> key(n_bins);
  for (int b = 0; b < n_bins; ++b) {
    const float c = hp[b * 3 + 2];
    key[b] = (c > 0.f) ? hp[b * 3] / (hp[b * 3 + 1] + smooth) : 1e30f;
    order[b] = b;
  }
  std::sort(order.begin(), order.end(), [&](int a, int b) {
    if (key[a] != key[b]) return key[a] < key[b];
    return a < b;
  });
}

void cpu_split_scan(const float* hist, const int32_t* abs_of_slot,
                    float* node_stats, float* best_gain_nf,
                    int32_t* best_bin_nf, int32_t* best_feat,
                    int32_t* best_bin, float* best_gain,
                    const uint8_t* feat_mask, const uint8_t* cat_flags,
                    unsigned long long* masks, const int8_t* mono,
                    float* node_bounds, int32_t* na_meanb_nf,
                    uint8_t* tree_na, int F, int n_bins, int slot0,
                    int n_slots, SplitParams sp) {
  ThreadPool::Get().ParallelFor(n_slots, [&](int slot) {
    std::vector<int> order;
    const int out = slot0 + slot;
    const int abs_node = abs_of_slot[out];
    // Node totals from feature 0 (every example lands in some bin).
    const float* h0 = hist + ((int64_t)slot * F + 0) * (n_bins * 3);
    float G = 0.f, H = 0.f, C = 0.f;
    for (int b = 0; b < n_bins; ++b) {
      G += h0[b * 3]; H += h0[b * 3 + 1]; C += h0[b * 3 + 2];
    }
    if (C > 0.f) {  // empty dense-mode slots must not zero parent-set stats
      float* ns = node_stats + (int64_t)abs_node * 3;
      ns[0] = G; ns[1] = H; ns[2] = C;
    }
    float node_best_gain = -1e30f;
    int node_best_f = -1, node_best_b = 0;
    for (int f = 0; f < F; ++f) {
      if (feat_mask != nullptr && !feat_mask[(int64_t)out * F + f]) {
        best_gain_nf[(int64_t)slot * F + f] = -1e30f;
        best_bin_nf[(int64_t)slot * F + f] = 0;
        continue;
      }
      const float* hp = hist + ((int64_t)slot * F + f) * (n_bins * 3);
      const bool is_cat = cat_flags != nullptr && cat_flags[f];
      if (is_cat) sorted_bin_order(hp, n_bins, sp.cat_smooth, order);
      auto bin_at = [&](int b) { return is_cat ? order[b] : b; };
      // LOCAL_IMPUTATION: fold the NA bin (255) into the node-local
      // mean bin before scanning (mirrors the GPU kernel)
      int meanb = -1;
      if (sp.na_mode && !is_cat) {
        float wsum = 0.f, csum = 0.f;
        for (int b = 0; b < n_bins - 1; ++b) {
          wsum += (float)b * hp[b * 3 + 2];
          csum += hp[b * 3 + 2];
        }
        meanb = csum > 0.f ? (int)(wsum / csum + 0.5f) : 0;
        if (meanb > n_bins - 2) meanb = n_bins - 2;
        if (na_meanb_nf != nullptr)
          na_meanb_nf[(int64_t)slot * F + f] = meanb;
      }
      auto hv = [&](int b, int j) -> float {
        float v = hp[b * 3 + j];
        if (meanb >= 0) {
          if (b == n_bins - 1) return 0.f;
          if (b == meanb) v += hp[(n_bins - 1) * 3 + j];
        }
        return v;
      };
      // Per-feature totals (matches the GPU kernel exactly; identical to
      // the feature-0 totals for any real histogram).
      float Gf = 0.f, Hf = 0.f, Cf = 0.f;
      for (int b = 0; b < n_bins; ++b) {
        Gf += hv(b, 0); Hf += hv(b, 1); Cf += hv(b, 2);
      }
      const float tpf = ydfa::l1_thresh(Gf, sp.lambda_l1);
      const float pterm = tpf * tpf / (Hf + sp.lambda_l2);
      float GL = 0.f, HL = 0.f, CL = 0.f;
      float fbest = -1e30f;
      int fbin = 0;
      for (int b = 0; b < n_bins - 1; ++b) {
        const int bb = bin_at(b);
        GL += hv(bb, 0); HL += hv(bb, 1); CL += hv(bb, 2);
        const float GR = Gf - GL, HR = Hf - HL, CR = Cf - CL;
        if (CL >= sp.min_examples && CR >= sp.min_examples &&
            HL >= sp.min_hessian && HR >= sp.min_hessian) {
          bool okm = true;
          if (mono != nullptr && mono[f] != 0) {
            const float wl = -ydfa::l1_thresh(GL, sp.lambda_l1)
                             / (HL + sp.lambda_l2);
            const float wr = -ydfa::l1_thresh(GR, sp.lambda_l1)
                             / (HR + sp.lambda_l2);
            okm = (mono[f] > 0) ? (wl <= wr) : (wl >= wr);
          }
          const float tl = ydfa::l1_thresh(GL, sp.lambda_l1);
          const float tr = ydfa::l1_thresh(GR, sp.lambda_l1);
          const float gain = tl * tl / (HL + sp.lambda_l2) +
                             tr * tr / (HR + sp.lambda_l2) - pterm;
          if (okm && gain > fbest) { fbest = gain; fbin = b; }
        }
      }
      best_gain_nf[(int64_t)slot * F + f] = fbest;
      best_bin_nf[(int64_t)slot * F + f] = fbin;
      if (fbest > node_best_gain) {
        node_best_gain = fbest;
        node_best_f = f;
        node_best_b = fbin;
      }
    }
    if (node_best_f < 0 || node_best_gain <= sp.min_gain) {
      best_feat[out] = -1;
      best_bin[out] = 0;
      best_gain[out] = 0.f;
      return;
    }
    best_feat[out] = node_best_f;
    best_bin[out] = node_best_b;
    best_gain[out] = node_best_gain;
    const float* hp = hist + ((int64_t)slot * F + node_best_f) * (n_bins * 3);
    const bool is_cat = cat_flags != nullptr && cat_flags[node_best_f];
    float GL = 0.f, HL = 0.f, CL = 0.f;
    if (is_cat) {
      sorted_bin_order(hp, n_bins, sp.cat_smooth, order);
      unsigned long long m[kMaxBins / 64] = {0};
      for (int r = 0; r < n_bins; ++r) {
        const int b = order[r];
        if (r <= node_best_b) {
          GL += hp[b * 3]; HL += hp[b * 3 + 1]; CL += hp[b * 3 + 2];
        } else if (hp[b * 3 + 2] > 0.f) {
          m[b >> 6] |= 1ull << (b & 63);  // goes RIGHT
        }
      }
      if (masks != nullptr)
        for (int w = 0; w < kMaxBins / 64; ++w)
          masks[(int64_t)abs_node * (kMaxBins / 64) + w] = m[w];
    } else {
      for (int b = 0; b <= node_best_b; ++b) {
        GL += hp[b * 3]; HL += hp[b * 3 + 1]; CL += hp[b * 3 + 2];
      }
      if (sp.na_mode && na_meanb_nf != nullptr) {
        const int mb = na_meanb_nf[(int64_t)slot * F + node_best_f];
        if (tree_na != nullptr)
          tree_na[abs_node] = mb > node_best_b ? (uint8_t)1 : (uint8_t)0;
        if (mb <= node_best_b) {
          GL += hp[(n_bins - 1) * 3];
          HL += hp[(n_bins - 1) * 3 + 1];
          CL += hp[(n_bins - 1) * 3 + 2];
        }
      }
    }
    float* nl = node_stats + (int64_t)(2 * abs_node + 1) * 3;
    float* nr = node_stats + (int64_t)(2 * abs_node + 2) * 3;
    nl[0] = GL; nl[1] = HL; nl[2] = CL;
    nr[0] = G - GL; nr[1] = H - HL; nr[2] = C - CL;
    if (node_bounds != nullptr) {
      const float lo = node_bounds[2 * abs_node];
      const float hi = node_bounds[2 * abs_node + 1];
      float llo = lo, lhi = hi, rlo = lo, rhi = hi;
      if (!is_cat && mono != nullptr && mono[node_best_f] != 0) {
        const float wl =
            std::min(std::max(-ydfa::l1_thresh(GL, sp.lambda_l1)
                              / (HL + sp.lambda_l2), lo), hi);
        const float wr = std::min(
            std::max(-ydfa::l1_thresh(G - GL, sp.lambda_l1)
                     / (H - HL + sp.lambda_l2), lo), hi);
        const float mid = 0.5f * (wl + wr);
        if (mono[node_best_f] > 0) { lhi = mid; rlo = mid; }
        else { llo = mid; rhi = mid; }
      }
      node_bounds[2 * (2 * abs_node + 1)] = llo;
      node_bounds[2 * (2 * abs_node + 1) + 1] = lhi;
      node_bounds[2 * (2 * abs_node + 2)] = rlo;
      node_bounds[2 * (2 * abs_node + 2) + 1] = rhi;
    }
  });
}

// Dense-mode planning (see plan_level_kernel in train_kernels.hip).
void cpu_plan_level(const float* node_stats, const int32_t* prev_best_feat,
                    int level_base, int level_size, int need, int use_sub,
                    int32_t* build_map, uint8_t* derived) {
  for (int rel = 0; rel < level_size; ++rel) {
    const int a = level_base + rel;
    const int split = prev_best_feat[rel >> 1] >= 0;
    const float ca = node_stats[(int64_t)a * 3 + 2];
    const float cs = node_stats[(int64_t)(level_base + (rel ^ 1)) * 3 + 2];
    const int elig_a = split && ca >= (float)need;
    const int elig_s = split && cs >= (float)need;
    const int is_right = (a & 1) == 0;
    const int dv = use_sub && elig_a && elig_s &&
                   (ca > cs || (ca == cs && is_right));
    derived[rel] = (uint8_t)dv;
    build_map[rel] = (elig_a && !dv) ? rel : -1;
  }
}

void cpu_subtract_hist(float* hist, const float* hist_prev,
                       const uint8_t* derived, int level_size, int F,
                       int n_bins) {
  const int64_t cells = (int64_t)F * n_bins * 3;
  ThreadPool::Get().ParallelFor(level_size, [&](int rel) {
    if (!derived[rel]) return;
    float* dst = hist + (int64_t)rel * cells;
    const float* par = hist_prev + (int64_t)(rel >> 1) * cells;
    const float* sib = hist + (int64_t)(rel ^ 1) * cells;
    for (int64_t k = 0; k < cells; ++k) dst[k] = par[k] - sib[k];
  });
}

void cpu_update_node_ids(const uint8_t* bins, int32_t* node_ids,
                         const int32_t* slot_map, const int32_t* best_feat,
                         const int32_t* best_bin, const uint8_t* cat_flags,
                         const unsigned long long* masks,
                         const uint8_t* tree_na, int64_t N,
                         int level_base, int level_size) {
  const int nb = std::max(1, std::min<int>(ThreadPool::Get().size(),
                                           (int)(N / 16384) + 1));
  const int64_t per = (N + nb - 1) / nb;
  ThreadPool::Get().ParallelFor(nb, [&](int blk) {
    const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
    for (int64_t k = i0; k < i1; ++k) {
      const int nid = node_ids[k];
      const int rel = nid - level_base;
      if (rel < 0 || rel >= level_size) continue;
      const int slot = slot_map[rel];
      if (slot < 0) continue;
      const int f = best_feat[slot];
      if (f < 0) continue;
      const int b = bins[(int64_t)f * N + k];
      int right;
      if (cat_flags != nullptr && cat_flags[f])
        right = (int)((masks[(int64_t)nid * (kMaxBins / 64) + (b >> 6)]
                       >> (b & 63)) & 1ull);
      else if (tree_na != nullptr && b == kMaxBins - 1)
        right = tree_na[nid];
      else
        right = b > best_bin[slot] ? 1 : 0;
      node_ids[k] = 2 * nid + 1 + right;
    }
  });
}

void cpu_leaf_values(const float* node_stats, const float* node_bounds,
                     float* leaf_values, int total_nodes, float lambda_l2,
                     float lambda_l1) {
  for (int i = 0; i < total_nodes; ++i) {
    const float* ns = node_stats + (int64_t)i * 3;
    float v = (ns[1] != 0.f)
                  ? (-ydfa::l1_thresh(ns[0], lambda_l1)
                     / (ns[1] + lambda_l2))
                  : 0.f;
    if (node_bounds != nullptr)
      v = std::min(std::max(v, node_bounds[2 * i]), node_bounds[2 * i + 1]);
    leaf_values[i] = v;
  }
}

void cpu_update_preds(float* preds, const int32_t* node_ids,
                      const float* leaf_values, int64_t N, float shrinkage) {
  const int nb = std::max(1, std::min<int>(ThreadPool::Get().size(),
                                           (int)(N / 16384) + 1));
  const int64_t per = (N + nb - 1) / nb;
  ThreadPool::Get().ParallelFor(nb, [&](int blk) {
    const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
    for (int64_t k = i0; k < i1; ++k) {
      const int nid = node_ids[k];
      if (nid >= 0) preds[k] += shrinkage * leaf_values[nid];
    }
  });
}

void cpu_binary_logloss(const float* preds, const float* labels, float* out2,
                        int64_t N) {
  double loss = 0.0, acc = 0.0;
  for (int64_t k = 0; k < N; ++k) {
    const float m = preds[k];
    const float y = labels[k];
    const float z = y > 0.5f ? -m : m;
    loss += (z > 0.f) ? z + std::log1p(std::exp(-z)) : std::log1p(std::exp(z));
    acc += ((m > 0.f) == (y > 0.5f)) ? 1.0 : 0.0;
  }
  out2[0] += (float)loss;
  out2[1] += (float)acc;
}

void cpu_predict_forest(const float* X, int64_t N, int F, const int32_t* feat,
                        const float* thr, const int32_t* left,
                        const int32_t* roots, const int32_t* cat_idx,
                        const unsigned long long* masks,
                        const int32_t* obl_ranges, const int32_t* obl_attr,
                        const float* obl_w, const uint8_t* na_right,
                        int tree_start,
                        int tree_step, int n_trees, float* out, float init,
                        float scale) {
  (void)F;
  const int nb = std::max(1, std::min<int>(ThreadPool::Get().size(),
                                           (int)(N / 1024) + 1));
  const int64_t per = (N + nb - 1) / nb;
  ThreadPool::Get().ParallelFor(nb, [&](int blk) {
    const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
    for (int64_t k = i0; k < i1; ++k) {
      float acc = init;
      for (int tt = 0; tt < n_trees; ++tt) {
        int n = roots[tree_start + (int64_t)tt * tree_step];
        int f = feat[n];
        while (f >= 0) {
          int right;
          const int ci = cat_idx ? cat_idx[n] : -1;
          const float xv = X[(int64_t)f * N + k];
          if (na_right != nullptr &&
              (std::isnan(xv) || (xv < 0.f && ci >= 0))) {
            right = na_right[n];  // missing input: stored na_value side
          } else if (ci >= 0) {
            int cbin = (int)xv;
            cbin = cbin < 0 ? 0 : (cbin > 255 ? 255 : cbin);
            right = (int)((masks[(int64_t)ci * 4 + (cbin >> 6)]
                           >> (cbin & 63)) & 1ull);
          } else if (ci <= -2) {  // oblique sparse projection
            const int oi = -(ci + 2);
            const int s0 = obl_ranges[2 * oi], nn = obl_ranges[2 * oi + 1];
            float dot = 0.f;
            for (int q = 0; q < nn; ++q)
              dot += obl_w[s0 + q] * X[(int64_t)obl_attr[s0 + q] * N + k];
            right = dot > thr[n] ? 1 : 0;
          } else {
            right = xv > thr[n] ? 1 : 0;
          }
          n = left[n] + right;
          f = feat[n];
        }
        acc += thr[n];
      }
      out[k] = init + (acc - init) * scale;
    }
  });
}

// ---------------------------------------------------------------------------
// TreeSHAP (Lundberg et al. 2018, consistent with reference utils/shap.h:83):
// path-dependent Shapley values for one flat-forest tree ensemble.
// phi layout: [N][F+1], last column = bias (expected value).
// ---------------------------------------------------------------------------
namespace {

struct PathElem {
  int feature;
  float zero_frac;
  float one_frac;
  float pweight;
};

void ShapExtend(PathElem* path, int depth, float pz, float po, int pi) {
  path[depth].feature = pi;
  path[depth].zero_frac = pz;
  path[depth].one_frac = po;
  path[depth].pweight = depth == 0 ? 1.0f : 0.0f;
  for (int i = depth - 1; i >= 0; --i) {
    path[i + 1].pweight += po * path[i].pweight * (i + 1) / (float)(depth + 1);
    path[i].pweight = pz * path[i].pweight * (depth - i) / (float)(depth + 1);
  }
}

void ShapUnwind(PathElem* path, int depth, int idx) {
  const float po = path[idx].one_frac;
  const float pz = path[idx].zero_frac;
  float next = path[depth].pweight;
  for (int i = depth - 1; i >= 0; --i) {
    if (po != 0.f) {
      const float tmp = path[i].pweight;
      path[i].pweight = next * (depth + 1) / ((i + 1) * po);
      next = tmp - path[i].pweight * pz * (depth - i) / (float)(depth + 1);
    } else {
      path[i].pweight =
          path[i].pweight * (depth + 1) / ((float)(depth - i) * pz);
    }
  }
  for (int i = idx; i < depth; ++i) {
    path[i].feature = path[i + 1].feature;
    path[i].zero_frac = path[i + 1].zero_frac;
    path[i].one_frac = path[i + 1].one_frac;
  }
}

float ShapUnwoundSum(const PathElem* path, int depth, int idx) {
  const float po = path[idx].one_frac;
  const float pz = path[idx].zero_frac;
  float total = 0.f;
  float next = path[depth].pweight;
  for (int i = depth - 1; i >= 0; --i) {
    if (po != 0.f) {
      const float tmp = next * (depth + 1) / ((i + 1) * po);
      total += tmp;
      next = path[i].pweight - tmp * pz * (depth - i) / (float)(depth + 1);
    } else {
      total += path[i].pweight / (pz * (depth - i) / (float)(depth + 1));
    }
  }
  return total;
}

struct ShapCtx {
  const float* X;
  int64_t N;
  const int32_t* feat;
  const float* thr;
  const int32_t* left;
  const int32_t* cat_idx;
  const unsigned long long* masks;
  const int32_t* obl_ranges;
  const int32_t* obl_attr;
  const float* obl_w;
  const uint8_t* na_right;
  const float* cover;
  double* phi;  // [F+1]
  int64_t row;
  float scale;
};

int ShapGoesRight(const ShapCtx& c, int node) {
  const int ci = c.cat_idx ? c.cat_idx[node] : -1;
  if (c.na_right != nullptr) {
    const float xv = c.X[(int64_t)c.feat[node] * c.N + c.row];
    if (std::isnan(xv) || (xv < 0.f && ci >= 0)) return c.na_right[node];
  }
  if (ci >= 0) {
    int cb = (int)c.X[(int64_t)c.feat[node] * c.N + c.row];
    cb = cb < 0 ? 0 : (cb > 255 ? 255 : cb);
    return (int)((c.masks[(int64_t)ci * 4 + (cb >> 6)] >> (cb & 63)) & 1ull);
  }
  if (ci <= -2) {  // oblique: credit goes to feat[node] (first attribute)
    const int oi = -(ci + 2);
    const int s0 = c.obl_ranges[2 * oi], nn = c.obl_ranges[2 * oi + 1];
    float dot = 0.f;
    for (int q = 0; q < nn; ++q)
      dot += c.obl_w[s0 + q] * c.X[(int64_t)c.obl_attr[s0 + q] * c.N + c.row];
    return dot > c.thr[node] ? 1 : 0;
  }
  return c.X[(int64_t)c.feat[node] * c.N + c.row] > c.thr[node] ? 1 : 0;
}

void ShapRecurse(const ShapCtx& c, int node, PathElem* parent_path,
                 int depth, float pz, float po, int pi) {
  PathElem path[64];
  for (int i = 0; i < depth; ++i) path[i] = parent_path[i];
  ShapExtend(path, depth, pz, po, pi);
  if (c.feat[node] < 0) {  // leaf
    for (int i = 1; i <= depth; ++i) {
      const float w = ShapUnwoundSum(path, depth, i);
      c.phi[path[i].feature] +=
          w * (path[i].one_frac - path[i].zero_frac) * c.thr[node] * c.scale;
    }
    return;
  }
  const int hot = c.left[node] + ShapGoesRight(c, node);
  const int cold = c.left[node] + (1 - ShapGoesRight(c, node));
  const float cover_n = c.cover[node] > 0.f ? c.cover[node] : 1.f;
  float hot_z = c.cover[hot] / cover_n;
  float cold_z = c.cover[cold] / cover_n;
  float incoming_z = 1.f, incoming_o = 1.f;
  int path_idx = -1;
  for (int i = 1; i <= depth; ++i) {
    if (path[i].feature == c.feat[node]) { path_idx = i; break; }
  }
  int new_depth = depth;
  if (path_idx >= 0) {
    incoming_z = path[path_idx].zero_frac;
    incoming_o = path[path_idx].one_frac;
    ShapUnwind(path, depth, path_idx);
    new_depth = depth - 1;
  }
  ShapRecurse(c, hot, path, new_depth + 1, hot_z * incoming_z, incoming_o,
              c.feat[node]);
  ShapRecurse(c, cold, path, new_depth + 1, cold_z * incoming_z, 0.f,
              c.feat[node]);
}

}  // namespace

extern "C" void cpu_tree_shap(const float* X, int64_t N, int F,
                              const int32_t* feat, const float* thr,
                              const int32_t* left, const int32_t* cat_idx,
                              const unsigned long long* masks,
                              const int32_t* obl_ranges,
                              const int32_t* obl_attr, const float* obl_w,
                              const uint8_t* na_right,
                              const float* cover, const int32_t* roots,
                              int tree_start, int tree_step, int n_trees,
                              float scale, float init, float* phi_out) {
  ThreadPool::Get().ParallelFor(
      std::max(1, std::min<int>(ThreadPool::Get().size() * 4,
                                (int)((N + 63) / 64))),
      [&](int blk) {
        const int nb = std::max(1, std::min<int>(
            ThreadPool::Get().size() * 4, (int)((N + 63) / 64)));
        const int64_t per = (N + nb - 1) / nb;
        const int64_t i0 = blk * per, i1 = std::min<int64_t>(i0 + per, N);
        std::vector<double> phi(F + 1);
        for (int64_t row = i0; row < i1; ++row) {
          std::fill(phi.begin(), phi.end(), 0.0);
          for (int tt = 0; tt < n_trees; ++tt) {
            const int root = roots[tree_start + (int64_t)tt * tree_step];
            ShapCtx c{X,          N,        feat,  thr,   left,
                      cat_idx,    masks,    obl_ranges, obl_attr, obl_w,
                      na_right,   cover,    phi.data(), row,  scale};
            PathElem dummy[1];
            ShapRecurse(c, root, dummy, 0, 1.f, 1.f, -1);
          }
          // bias column filled by the Python wrapper (init + E[forest])
          phi[F] = 0.0;
          (void)init;
          float* out = phi_out + (int64_t)row * (F + 1);
          for (int k = 0; k <= F; ++k) out[k] = (float)phi[k];
        }
      });
}

extern "C" void cpu_forest_expected_value(
    const int32_t* feat, const float* thr, const int32_t* left,
    const float* cover, const int32_t* roots, int tree_start, int tree_step,
    int n_trees, float scale, double* out) {
  double total = 0.0;
  for (int tt = 0; tt < n_trees; ++tt) {
    // iterative cover-weighted mean of leaf values
    const int root = roots[tree_start + (int64_t)tt * tree_step];
    struct Item { int node; double w; };
    std::vector<Item> stack{{root, 1.0}};
    double ev = 0.0;
    while (!stack.empty()) {
      Item it = stack.back();
      stack.pop_back();
      if (feat[it.node] < 0) {
        ev += it.w * thr[it.node];
        continue;
      }
      const int l = left[it.node];
      const double cn = cover[it.node] > 0 ? cover[it.node] : 1.0;
      stack.push_back({l, it.w * cover[l] / cn});
      stack.push_back({l + 1, it.w * cover[l + 1] / cn});
    }
    total += ev * scale;
  }
  *out = total;
}

}  // extern "C"
}  // namespace ydfa
