// Common definitions shared by the CPU and GPU (HIP/gfx950) implementations
// of the decision-forest training/inference ops.
//
// Design (MI355X-native, not a port):
//   * Features are pre-binned to uint8 (<=256 quantile bins), stored
//     feature-major: bins[f * N + i]. This is the GPU-resident analogue of the
//     reference's DISCRETIZED_NUMERICAL columns + dataset_cache binned store
//     (reference: yggdrasil_decision_forests/learner/distributed_decision_tree/
//     dataset_cache/dataset_cache.h:15-58).
//   * Trees are grown level-wise ("open nodes" of one depth at a time, like
//     the reference's distributed layer-wise growth, training.h:145) over an
//     implicit complete binary tree: node k has children 2k+1 / 2k+2.
//   * Per-level histograms hist[node][feature][bin] = {sum_grad, sum_hess,
//     count} are the collective unit: data-parallel ranks AllReduce this
//     tensor (RCCL over xGMI) and then select splits redundantly.
#pragma once
#include <cmath>
#include <cstdint>

// Host-only builds (the C++ user API links cpu_ops.cpp with a plain
// C++ compiler): the HIP function-space qualifiers become no-ops.
#ifndef __HIPCC__
#ifndef __host__
#define __host__
#endif
#ifndef __device__
#define __device__
#endif
#endif

namespace ydfa {

// Fixed maximum number of bins. Runtime bin count n_bins <= kMaxBins.
constexpr int kMaxBins = 256;

// Loss ids (subset of reference model/gradient_boosted_trees/
// gradient_boosted_trees.proto:54-80 enum).
enum LossKind : int {
  kLossSquaredError = 2,      // regression: g = pred - y, h = 1
  kLossBinomial = 1,          // binary classification log-loss on logits
  kLossMultinomial = 3,       // multi-class softmax cross-entropy
  kLossPoisson = 7,           // log-link Poisson: g = exp(p) - y, h = exp(p)
  kLossMAE = 8,               // mean absolute error: g = sign(p - y), h = 1
};

// Split-scan hyper-parameters (subset of the reference decision-tree proto).
struct SplitParams {
  float lambda_l2;        // l2_regularization
  float lambda_l1;        // l1_regularization (soft-thresholds gradients,
                          // XGBoost eq. 2 formulation)
  float min_hessian;      // min_sum_hessian_in_leaf
  int min_examples;       // min_examples (reference default 5)
  float min_gain;         // splits with gain <= min_gain become leaves
  float cat_smooth;       // l2_categorical_regularization (category order
                          // statistic smoothing; reference default 1.0)
  int na_mode;            // LOCAL_IMPUTATION: bin 255 holds NaN rows;
                          // the scan merges it into the node-local mean
                          // bin and records the na direction
};

// l1 soft threshold: T(G) = sign(G) * max(|G| - l1, 0)
__host__ __device__ inline float l1_thresh(float g, float l1) {
  const float a = fabsf(g) - l1;
  return a > 0.f ? (g > 0.f ? a : -a) : 0.f;
}

}  // namespace ydfa
