// HIP/CDNA4 (gfx950) training kernels for the MI355X-native decision-forest
// engine. Hand-written for 64-wide wavefronts, LDS-staged histograms and
// HBM3E-friendly access patterns — a new design, not a translation of the
// reference's CPU splitter (SURVEY.md §2.3 maps reference hot paths to these
// kernels).
//
// Level-wise growth over an implicit complete binary tree (node k ->
// children 2k+1 / 2k+2), with a SPARSE active-node list per level: the host
// keeps the (allreduce-deterministic) list of open nodes; `slot_map` maps a
// node's level-relative index to its dense histogram slot (-1 = closed).
// This keeps hist memory O(open nodes), not O(2^depth), so depth-16 random
// forests stay feasible.
//
// Pipeline per tree (host loop in ydf_amd/learner/trainer.py):
//   grad_hess / weighted_target : per-example {g,h}
//                      (ref: loss_imp_*.cc UpdateGradients, elementwise)
//   per level L, per slot-chunk:
//     hist_build     : hist[slot][feat][bin] = {sum_g, sum_h, count}
//                      (ref: splitter_scanner.h:95-185 bucket fill)
//     [multi-GPU: RCCL AllReduce(hist) — inserted by the Python host loop]
//     split_scan     : prefix-scan bins + deterministic argmax gain per slot
//                      (ref: splitter_scanner.h:933-1101 ScanSplits)
//   update_node_ids  : route examples to children (ref: training.cc:5324
//                      SplitExamplesInPlace — here a node-id map update;
//                      rows never move)
//   leaf_values + update_preds
//                      (ref: gradient_boosted_trees.cc:1576 UpdatePredictions)
#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include "common.h"

namespace ydfa {

constexpr int kBlock = 256;

// ---------------------------------------------------------------------------
// Binning: x[f*N+i] -> bin index via upper-bound binary search over the
// feature's quantile boundaries (boundaries[f*n_cuts .. +n_cuts), ascending).
// bin = number of cuts strictly below v, so "bin > b" <=> "v > cut[b]".
// Boundaries staged in LDS (<= 255 floats per feature).
// ---------------------------------------------------------------------------
__global__ void bin_data_kernel(const float* __restrict__ x,
                                const float* __restrict__ boundaries,
                                uint8_t* __restrict__ out, int64_t N, int F,
                                int n_cuts, int64_t rows_per_block,
                                int na_to_255) {
  __shared__ float bnd[kMaxBins - 1];
  const int f = blockIdx.x;
  for (int i = threadIdx.x; i < n_cuts; i += blockDim.x)
    bnd[i] = boundaries[(int64_t)f * n_cuts + i];
  __syncthreads();
  const int64_t row0 = (int64_t)blockIdx.y * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, N);
  const float* xf = x + (int64_t)f * N;
  uint8_t* of = out + (int64_t)f * N;
  for (int64_t i = row0 + threadIdx.x; i < row1; i += blockDim.x) {
    const float v = xf[i];
    if (na_to_255 && isnan(v)) {
      of[i] = 255;  // reserved NA bin (LOCAL_IMPUTATION)
      continue;
    }
    int lo = 0, hi = n_cuts;
    while (lo < hi) {
      const int mid = (lo + hi) >> 1;
      if (bnd[mid] < v) lo = mid + 1; else hi = mid;
    }
    of[i] = (uint8_t)lo;
  }
}

// ---------------------------------------------------------------------------
// Gradients/hessians, written interleaved as float2 {g, h} for single-load
// consumption by the histogram kernel.
// ---------------------------------------------------------------------------
__global__ void grad_hess_kernel(const float* __restrict__ preds,
                                 const float* __restrict__ labels,
                                 float2* __restrict__ gh, int64_t N,
                                 int loss) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    float g, h;
    if (loss == kLossBinomial) {
      const float p = 1.0f / (1.0f + __expf(-preds[k]));
      g = p - labels[k];
      h = fmaxf(p * (1.0f - p), 1e-16f);
    } else if (loss == kLossPoisson) {
      const float ep = __expf(fminf(preds[k], 15.0f));
      g = ep - labels[k];
      h = fmaxf(ep, 1e-6f);
    } else if (loss == kLossMAE) {
      g = (preds[k] > labels[k]) ? 1.0f : -1.0f;
      h = 1.0f;
    } else {  // squared error
      g = preds[k] - labels[k];
      h = 1.0f;
    }
    gh[k] = make_float2(g, h);
  }
}

// Multi-class softmax cross-entropy: preds [C][N] (class-major), labels are
// class indices. Writes gh for ONE class `cls` (host loops classes when
// building the per-class trees of one iteration, matching the reference's
// one-tree-per-class MULTINOMIAL loop, gradient_boosted_trees.cc:1539).
__global__ void grad_hess_softmax_kernel(const float* __restrict__ preds,
                                         const float* __restrict__ labels,
                                         float2* __restrict__ gh, int64_t N,
                                         int n_classes, int cls) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    float m = -1e30f;
    for (int c = 0; c < n_classes; ++c)
      m = fmaxf(m, preds[(int64_t)c * N + k]);
    float denom = 0.f;
    for (int c = 0; c < n_classes; ++c)
      denom += __expf(preds[(int64_t)c * N + k] - m);
    const float p = __expf(preds[(int64_t)cls * N + k] - m) / denom;
    const float y = (labels[k] == (float)cls) ? 1.0f : 0.0f;
    gh[k] = make_float2(p - y, fmaxf(p * (1.0f - p), 1e-16f));
  }
}

// Random-forest / CART target: g = -w*y, h = w (leaf value -G/H = weighted
// mean of y; gain = weighted variance reduction, which on 0/1 labels orders
// splits like Gini). `weights` may be null (unit weights); bootstrap
// sampling passes per-example draw counts as weights (w=0 = out-of-bag).
__global__ void weighted_target_kernel(const float* __restrict__ labels,
                                       const float* __restrict__ weights,
                                       float2* __restrict__ gh, int64_t N) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    const float w = weights ? weights[k] : 1.0f;
    gh[k] = make_float2(-w * labels[k], w);
  }
}

// ---------------------------------------------------------------------------
// Histogram build: one block owns (feature f, row chunk) and accumulates a
// private LDS histogram for a CONTIGUOUS RANGE of slots [slot0, slot0+ng)
// (ng * n_bins * 12 B <= 160 KiB => ng <= 53 at 256 bins; the host launcher
// slices bigger levels). Examples resolve level-relative node -> slot via
// slot_map; closed nodes (slot -1) and out-of-level examples are skipped.
// The count accumulates examples with h != 0 so that zero-weight (out-of-
// bag) rows don't satisfy min_examples.
// ---------------------------------------------------------------------------
// Inner-loop design, measured on MI355X (tools/kernel_lab.hip):
//   * 32-bit LDS atomics are the bottleneck of the classic 3x ds_add_f32
//     histogram (~205 G atomics/s; the LDS array sits ~100% busy), while
//     64-bit LDS atomics (ds_add_f64 / ds_add_u64) run ~5x faster per op
//     (~955 G/s, at the load floor). So each bin accumulates with exactly
//     TWO 64-bit atomics: g in f64 (exact), and {h fixed-point * 2^20,
//     count << 44} packed in one u64 — 690 Gvisits/s vs 68 for the f32
//     form (10x). Constraint: per-example h <= 16 and rows_per_block <=
//     2^19 keep the h field below 2^44 (trainer clips bootstrap weights).
//   * slot_map staged in LDS when the level fits (lds_map != 0), removing a
//     dependent global load per (row, feature) visit;
//   * rows processed kHistUnroll at a time with UNCONDITIONAL node/gh/bin
//     loads issued back-to-back (gh/bins are always-valid full arrays), so
//     the wave has kHistUnroll*3 loads in flight instead of 1 (the naive
//     branchy form serializes three dependent round-trips per row).
// The merge unpacks to the f32 {sum_g, sum_h, count} global tensor the
// scan/allreduce layers consume, skipping untouched bins.
constexpr int kHistUnroll = 4;
constexpr float kHScale = 1048576.0f;      // 2^20
constexpr float kHInvScale = 1.0f / 1048576.0f;
constexpr unsigned long long kHMask = (1ull << 44) - 1;

__global__ void hist_build_lds_kernel(const uint8_t* __restrict__ bins,
                                      const float2* __restrict__ gh,
                                      const int32_t* __restrict__ node_ids,
                                      const int32_t* __restrict__ slot_map,
                                      float* __restrict__ hist, int64_t N,
                                      int F, int n_bins, int level_base,
                                      int level_size, int slot0, int n_slots,
                                      const uint8_t* __restrict__ row_grp,
                                      int n_slots_group,
                                      int lds_map, int filtered, int fpb,
                                      int n_fgroups, int n_chunks,
                                      int swizzle, int64_t rows_per_block) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // carve: [fpb][n_slots][n_bins] {f64 g, u64 h|count}, then the slot map.
  // fpb > 1 ("multi-feature blocks") amortizes the gh/node_ids streams:
  // one pass over the rows feeds fpb features' histograms.
  double* lg = reinterpret_cast<double*>(smem);
  unsigned long long* lp =
      reinterpret_cast<unsigned long long*>(smem) + 1;  // interleaved pairs
  const int tot = fpb * n_slots * n_bins;
  int* lmap = reinterpret_cast<int*>(smem + (size_t)tot * 16);
  // XCD-aware mapping (cdna_hip_programming.md T1): the dispatcher places
  // block b on XCD b%8, so id%8 selects the row chunk and consecutive ids
  // on one XCD sweep the feature groups of that chunk — the chunk's gh and
  // node_ids stay hot in that XCD's private L2 (placement heuristic:
  // affects speed only, never correctness).
  int fgroup, chunk;
  if (swizzle) {
    const int id = blockIdx.x;
    const int xcd = id & 7;
    const int sid = id >> 3;
    chunk = xcd + 8 * (sid / n_fgroups);
    fgroup = sid - (sid / n_fgroups) * n_fgroups;
    if (chunk >= n_chunks) return;
  } else {
    fgroup = blockIdx.x;
    chunk = blockIdx.y;
  }
  const int f0 = fgroup * fpb;
  const int nf = (F - f0) < fpb ? (F - f0) : fpb;
  {
    unsigned long long* z = reinterpret_cast<unsigned long long*>(smem);
    for (int i = threadIdx.x; i < tot * 2; i += blockDim.x) z[i] = 0ull;
  }
  if (lds_map) {
    for (int i = threadIdx.x; i < level_size; i += blockDim.x)
      lmap[i] = slot_map[i];
  }
  __syncthreads();
  const int64_t row0 = (int64_t)chunk * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, N);
  const int64_t stride = blockDim.x;
  int64_t i = row0 + threadIdx.x;
  const int64_t bulk_end = row1 - (kHistUnroll - 1) * stride;

  if (filtered == 2 && row_grp != nullptr) {
    // slot8 mode: row_grp holds the per-row GLOBAL slot id precomputed
    // once per level (255 = row not built this level, e.g. the derived
    // sibling under histogram subtraction, or subsampled out). Streams
    // 1 B/row for the filter instead of node_ids (4 B) + slot-map
    // lookup; gh/bins load only under the match predicate.
    for (; i < bulk_end; i += kHistUnroll * stride) {
      uint8_t gb[kHistUnroll];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) gb[u] = row_grp[i + u * stride];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) {
        const int slot = (int)gb[u] - slot0;
        if (slot < 0 || slot >= n_slots) continue;
        const float2 v = gh[i + u * stride];
        const unsigned long long hq =
            (unsigned long long)(v.y * kHScale + 0.5f);
        const unsigned long long pk =
            hq | ((unsigned long long)(v.y != 0.f) << 44);
        for (int j = 0; j < nf; ++j) {
          const int b = bins[(int64_t)(f0 + j) * N + i + u * stride];
          const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
          atomicAdd(lg + cell, (double)v.x);
          atomicAdd(lp + cell, pk);
        }
      }
    }
    for (; i < row1; i += stride) {
      const int slot = (int)row_grp[i] - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      const float2 v = gh[i];
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f);
      const unsigned long long pk =
          hq | ((unsigned long long)(v.y != 0.f) << 44);
      for (int j = 0; j < nf; ++j) {
        const int b = bins[(int64_t)(f0 + j) * N + i];
        const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
        atomicAdd(lg + cell, (double)v.x);
        atomicAdd(lp + cell, pk);
      }
    }
  } else if (filtered && row_grp != nullptr) {
    // Cheapest multi-group pass: 1-byte group id per row; only matching
    // rows touch node_ids/gh/bins.
    const uint8_t my_grp = (uint8_t)(slot0 / n_slots_group);
    for (; i < bulk_end; i += kHistUnroll * stride) {
      uint8_t gb[kHistUnroll];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) gb[u] = row_grp[i + u * stride];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) {
        if (gb[u] != my_grp) continue;
        const int rel = node_ids[i + u * stride] - level_base;
        const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
        if (slot < 0 || slot >= n_slots) continue;
        const float2 v = gh[i + u * stride];
        const unsigned long long hq =
            (unsigned long long)(v.y * kHScale + 0.5f);
        const unsigned long long pk =
            hq | ((unsigned long long)(v.y != 0.f) << 44);
        for (int j = 0; j < nf; ++j) {
          const int b = bins[(int64_t)(f0 + j) * N + i + u * stride];
          const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
          atomicAdd(lg + cell, (double)v.x);
          atomicAdd(lp + cell, pk);
        }
      }
    }
    for (; i < row1; i += stride) {
      if (row_grp[i] != my_grp) continue;
      const int rel = node_ids[i] - level_base;
      const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      const float2 v = gh[i];
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f);
      const unsigned long long pk =
          hq | ((unsigned long long)(v.y != 0.f) << 44);
      for (int j = 0; j < nf; ++j) {
        const int b = bins[(int64_t)(f0 + j) * N + i];
        const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
        atomicAdd(lg + cell, (double)v.x);
        atomicAdd(lp + cell, pk);
      }
    }
  } else if (filtered) {
    // Multi-slot-group launch: most rows belong to ANOTHER group, so load
    // only node_ids eagerly and fetch gh/bins under the match predicate —
    // a group pass then streams ~4B/row instead of 13B/row.
    for (; i < bulk_end; i += kHistUnroll * stride) {
      int nid[kHistUnroll];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u)
        nid[u] = node_ids[i + u * stride];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) {
        const int rel = nid[u] - level_base;
        if (rel < 0 || rel >= level_size) continue;
        const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
        if (slot < 0 || slot >= n_slots) continue;
        const float2 v = gh[i + u * stride];
        const unsigned long long hq =
            (unsigned long long)(v.y * kHScale + 0.5f);
        const unsigned long long pk =
            hq | ((unsigned long long)(v.y != 0.f) << 44);
        for (int j = 0; j < nf; ++j) {
          const int b = bins[(int64_t)(f0 + j) * N + i + u * stride];
          const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
          atomicAdd(lg + cell, (double)v.x);
          atomicAdd(lp + cell, pk);
        }
      }
    }
    for (; i < row1; i += stride) {
      const int rel = node_ids[i] - level_base;
      if (rel < 0 || rel >= level_size) continue;
      const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      const float2 v = gh[i];
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f);
      const unsigned long long pk =
          hq | ((unsigned long long)(v.y != 0.f) << 44);
      for (int j = 0; j < nf; ++j) {
        const int b = bins[(int64_t)(f0 + j) * N + i];
        const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
        atomicAdd(lg + cell, (double)v.x);
        atomicAdd(lp + cell, pk);
      }
    }
  } else {
    for (; i < bulk_end; i += kHistUnroll * stride) {
      int nid[kHistUnroll];
      float2 v[kHistUnroll];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u)
        nid[u] = node_ids[i + u * stride];
#pragma unroll
      for (int u = 0; u < kHistUnroll; ++u) v[u] = gh[i + u * stride];
      for (int j = 0; j < nf; ++j) {
        const uint8_t* fb = bins + (int64_t)(f0 + j) * N;
        uint8_t b[kHistUnroll];
#pragma unroll
        for (int u = 0; u < kHistUnroll; ++u) b[u] = fb[i + u * stride];
#pragma unroll
        for (int u = 0; u < kHistUnroll; ++u) {
          const int rel = nid[u] - level_base;
          if (rel < 0 || rel >= level_size) continue;
          const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
          if (slot < 0 || slot >= n_slots) continue;
          const int cell = 2 * (((j * n_slots) + slot) * n_bins + (int)b[u]);
          atomicAdd(lg + cell, (double)v[u].x);
          const unsigned long long hq =
              (unsigned long long)(v[u].y * kHScale + 0.5f);
          atomicAdd(lp + cell,
                    hq | ((unsigned long long)(v[u].y != 0.f) << 44));
        }
      }
    }
    for (; i < row1; i += stride) {
      const int rel = node_ids[i] - level_base;
      if (rel < 0 || rel >= level_size) continue;
      const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      const float2 v = gh[i];
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f);
      const unsigned long long pk =
          hq | ((unsigned long long)(v.y != 0.f) << 44);
      for (int j = 0; j < nf; ++j) {
        const int b = bins[(int64_t)(f0 + j) * N + i];
        const int cell = 2 * (((j * n_slots) + slot) * n_bins + b);
        atomicAdd(lg + cell, (double)v.x);
        atomicAdd(lp + cell, pk);
      }
    }
  }
  __syncthreads();
  for (int k = threadIdx.x; k < tot; k += blockDim.x) {
    const double g = lg[2 * k];
    const unsigned long long pk = lp[2 * k];
    if (pk == 0ull && g == 0.0) continue;  // untouched bin
    const int j = k / (n_slots * n_bins);
    const int rem = k - j * (n_slots * n_bins);
    const int slot = rem / n_bins;
    const int bin = rem - slot * n_bins;
    if (f0 + j >= F) continue;
    float* p = hist + ((int64_t)slot * F + (f0 + j)) * (n_bins * 3) + bin * 3;
    atomicAdd(p, (float)g);
    // double-precision unpack: one rounding to f32 (keeps integer-valued
    // h sums exact, which the CPU/GPU equality tests rely on)
    atomicAdd(p + 1, (float)((double)(pk & kHMask) * (double)kHInvScale));
    atomicAdd(p + 2, (float)(pk >> 44));
  }
}

// Packs the per-tree host-extraction arrays into one contiguous byte
// buffer in a SINGLE launch (the torch slice-copy form costs 5-7
// separate ~5 us copyBuffer ops on sub-KB arrays). Layout mirrors
// trainer._extract_batched: feat[T] i32 | bin[T] i32 | leaf[T] f32 |
// counts[T] f32 (node_stats[i*3+2]) | gain[T] f32 | masks u64[W]
// (optional) | na u8[T] (optional).
__global__ void pack_extract_kernel(const int32_t* __restrict__ feat,
                                    const int32_t* __restrict__ binv,
                                    const float* __restrict__ leaf,
                                    const float* __restrict__ node_stats,
                                    const float* __restrict__ gain,
                                    const unsigned long long* __restrict__
                                        tmasks,
                                    const uint8_t* __restrict__ na,
                                    uint8_t* __restrict__ out, int T,
                                    int64_t n_mask_words) {
  int32_t* o_feat = reinterpret_cast<int32_t*>(out);
  int32_t* o_bin = o_feat + T;
  float* o_leaf = reinterpret_cast<float*>(o_bin + T);
  float* o_cnt = o_leaf + T;
  float* o_gain = o_cnt + T;
  // masks start 8-byte aligned (T is usually odd: 2^d - 1 nodes)
  const int64_t moff = ((int64_t)20 * T + 7) & ~7LL;
  unsigned long long* o_mask =
      reinterpret_cast<unsigned long long*>(out + moff);
  uint8_t* o_na = out + moff + 8 * n_mask_words;
  const int64_t n = T > n_mask_words ? T : n_mask_words;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    if (i < T) {
      o_feat[i] = feat[i];
      o_bin[i] = binv[i];
      o_leaf[i] = leaf[i];
      o_cnt[i] = node_stats[i * 3 + 2];
      o_gain[i] = gain[i];
      if (na != nullptr) o_na[i] = na[i];
    }
    if (tmasks != nullptr && i < n_mask_words) o_mask[i] = tmasks[i];
  }
}

// Precomputes the per-row slot-GROUP id (u8) for multi-group levels so each
// group pass streams 1 B/row instead of node_id + slot-map lookups.
// 255 = row not in any open slot of this level.
__global__ void row_group_kernel(const int32_t* __restrict__ node_ids,
                                 const int32_t* __restrict__ slot_map,
                                 uint8_t* __restrict__ grp, int64_t N,
                                 int level_base, int level_size,
                                 int group_size) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    const int rel = node_ids[k] - level_base;
    int g = 255;
    if (rel >= 0 && rel < level_size) {
      const int slot = slot_map[rel];
      if (slot >= 0) g = slot / group_size;
    }
    grp[k] = (uint8_t)g;
  }
}

// Gathered histogram build for DEEP levels (row partitioning): rows are
// pre-sorted by slot (torch stable argsort in the trainer), so one slot
// GROUP's rows form a contiguous range [row_lo, row_hi) of `row_order`.
// A group pass then touches only its own rows — at 1000+ open nodes this
// replaces ~30 full-table scans per level with one partitioned sweep
// (gh/node_ids are L2-resident; the scattered bins gathers are the cost).
// ---------------------------------------------------------------------------
__global__ void hist_build_gathered_kernel(
    const uint8_t* __restrict__ bins, const float2* __restrict__ gh,
    const int32_t* __restrict__ node_ids,
    const int32_t* __restrict__ slot_map,
    const int32_t* __restrict__ row_order, float* __restrict__ hist,
    int64_t N, int F, int n_bins, int level_base, int level_size, int slot0,
    int n_slots, int lds_map, int64_t row_lo, int64_t row_hi,
    int64_t rows_per_block) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* lg = reinterpret_cast<double*>(smem);
  unsigned long long* lp =
      reinterpret_cast<unsigned long long*>(smem) + 1;
  const int tot = n_slots * n_bins;
  int* lmap = reinterpret_cast<int*>(smem + (size_t)tot * 16);
  const int f = blockIdx.x;
  {
    unsigned long long* z = reinterpret_cast<unsigned long long*>(smem);
    for (int i = threadIdx.x; i < tot * 2; i += blockDim.x) z[i] = 0ull;
  }
  if (lds_map) {
    for (int i = threadIdx.x; i < level_size; i += blockDim.x)
      lmap[i] = slot_map[i];
  }
  __syncthreads();
  const uint8_t* fb = bins + (int64_t)f * N;
  const int64_t j0 = row_lo + (int64_t)blockIdx.y * rows_per_block;
  const int64_t j1 = min(j0 + rows_per_block, row_hi);
  const int64_t stride = blockDim.x;
  int64_t j = j0 + threadIdx.x;
  const int64_t bulk_end = j1 - (kHistUnroll - 1) * stride;
  for (; j < bulk_end; j += kHistUnroll * stride) {
    int rows[kHistUnroll];
    int nid[kHistUnroll];
#pragma unroll
    for (int u = 0; u < kHistUnroll; ++u)
      rows[u] = row_order[j + u * stride];
#pragma unroll
    for (int u = 0; u < kHistUnroll; ++u) nid[u] = node_ids[rows[u]];
#pragma unroll
    for (int u = 0; u < kHistUnroll; ++u) {
      const int rel = nid[u] - level_base;
      if (rel < 0 || rel >= level_size) continue;
      const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
      if (slot < 0 || slot >= n_slots) continue;
      const float2 v = gh[rows[u]];
      const int cell = 2 * (slot * n_bins + (int)fb[rows[u]]);
      atomicAdd(lg + cell, (double)v.x);
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f);
      atomicAdd(lp + cell,
                hq | ((unsigned long long)(v.y != 0.f) << 44));
    }
  }
  for (; j < j1; j += stride) {
    const int row = row_order[j];
    const int rel = node_ids[row] - level_base;
    if (rel < 0 || rel >= level_size) continue;
    const int slot = (lds_map ? lmap[rel] : slot_map[rel]) - slot0;
    if (slot < 0 || slot >= n_slots) continue;
    const float2 v = gh[row];
    const int cell = 2 * (slot * n_bins + (int)fb[row]);
    atomicAdd(lg + cell, (double)v.x);
    const unsigned long long hq =
        (unsigned long long)(v.y * kHScale + 0.5f);
    atomicAdd(lp + cell, hq | ((unsigned long long)(v.y != 0.f) << 44));
  }
  __syncthreads();
  for (int k = threadIdx.x; k < tot; k += blockDim.x) {
    const double g = lg[2 * k];
    const unsigned long long pk = lp[2 * k];
    if (pk == 0ull && g == 0.0) continue;
    const int slot = k / n_bins;
    const int bin = k - slot * n_bins;
    float* p = hist + ((int64_t)slot * F + f) * (n_bins * 3) + bin * 3;
    atomicAdd(p, (float)g);
    atomicAdd(p + 1, (float)((double)(pk & kHMask) * (double)kHInvScale));
    atomicAdd(p + 2, (float)(pk >> 44));
  }
}

// ---------------------------------------------------------------------------
// Row partition scatter: counting-sort pass replacing the generic
// device merge sort (keys are small slot ids). cursor[k] starts at the
// slot's exclusive-prefix offset; arrival order within a slot is
// arbitrary (histogram accumulation is order-insensitive up to ulps).
// ---------------------------------------------------------------------------
__global__ void row_scatter_kernel(const int32_t* __restrict__ keys,
                                   int32_t* __restrict__ cursor,
                                   int32_t* __restrict__ row_order,
                                   int64_t N) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t r = i; r < N; r += stride) {
    const int pos = atomicAdd(cursor + keys[r], 1);
    row_order[pos] = (int32_t)r;
  }
}

// Few-key variant: global atomics on a handful of counters serialize
// (measured 2.3 ms/call at shallow levels), so each block counts its
// contiguous row chunk in LDS, reserves one global range per key, then
// places rows through fast LDS cursors.
__global__ void row_scatter_block_kernel(const int32_t* __restrict__ keys,
                                         int32_t* __restrict__ cursor,
                                         int32_t* __restrict__ row_order,
                                         int64_t N, int n_keys) {
  extern __shared__ int lcnt[];
  for (int k = threadIdx.x; k < n_keys; k += blockDim.x) lcnt[k] = 0;
  __syncthreads();
  const int64_t per = (N + gridDim.x - 1) / gridDim.x;
  const int64_t c0 = (int64_t)blockIdx.x * per;
  const int64_t c1 = min(c0 + per, N);
  for (int64_t r = c0 + threadIdx.x; r < c1; r += blockDim.x)
    atomicAdd(lcnt + keys[r], 1);
  __syncthreads();
  for (int k = threadIdx.x; k < n_keys; k += blockDim.x) {
    const int c = lcnt[k];
    lcnt[k] = c ? atomicAdd(cursor + k, c) : 0;
  }
  __syncthreads();
  for (int64_t r = c0 + threadIdx.x; r < c1; r += blockDim.x) {
    const int pos = atomicAdd(lcnt + keys[r], 1);
    row_order[pos] = (int32_t)r;
  }
}

// ---------------------------------------------------------------------------
// 32-feature interleave variant: bins32 is [ceil(F/32)][N][32] u8, one
// row costs two uint4 loads from the SAME 32-byte line (vs two separate
// scattered lines with the 16-wide layout). One slot per block;
// LDS = 32 * 256 * 16 B = 128 KiB.
// ---------------------------------------------------------------------------
__global__ void hist_build_gathered32_kernel(
    const uint8_t* __restrict__ bins32, const float2* __restrict__ gh,
    const int32_t* __restrict__ node_ids,
    const int32_t* __restrict__ slot_map,
    const int32_t* __restrict__ row_order,
    const int64_t* __restrict__ group_offs, float* __restrict__ hist,
    const uint32_t* __restrict__ maskbits,  // [ns][F32] or null
    int64_t N, int F, int n_bins, int level_base, int level_size,
    int win0, int n_chunks) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* lg = reinterpret_cast<double*>(smem);
  unsigned long long* lp =
      reinterpret_cast<unsigned long long*>(smem) + 1;
  const int fg = blockIdx.x;
  const int slot_abs = blockIdx.z;   // one slot per block
  const int tot = 32 * n_bins;
  {
    unsigned long long* zz = reinterpret_cast<unsigned long long*>(smem);
    for (int i = threadIdx.x; i < tot * 2; i += blockDim.x) zz[i] = 0ull;
  }
  __syncthreads();
  const int64_t r0 = group_offs[slot_abs];
  const int64_t r1 = group_offs[slot_abs + 1];
  const int64_t per = (r1 - r0 + n_chunks - 1) / n_chunks;
  const int64_t j0 = r0 + (int64_t)blockIdx.y * per;
  const int64_t j1 = min(j0 + per, r1);
  const uint4* fb =
      reinterpret_cast<const uint4*>(bins32 + (int64_t)fg * N * 32);
  const int F32 = (F + 31) / 32;
  unsigned m = 0xFFFFFFFFu;
  if (maskbits != nullptr)
    m = maskbits[(int64_t)slot_abs * F32 + fg];
  if (m != 0u) {
    for (int64_t j = j0 + threadIdx.x; j < j1; j += blockDim.x) {
      const int row = row_order[j];
      const int rel = node_ids[row] - level_base;
      if (rel < 0 || rel >= level_size) continue;
      if (slot_map[rel] - win0 != slot_abs) continue;
      const float2 v = gh[row];
      const unsigned long long hq =
          (unsigned long long)(v.y * kHScale + 0.5f) |
          ((unsigned long long)(v.y != 0.f) << 44);
      const uint4 a = fb[(int64_t)row * 2];
      const uint4 b = fb[(int64_t)row * 2 + 1];
      const unsigned words[8] = {a.x, a.y, a.z, a.w, b.x, b.y, b.z, b.w};
      unsigned mm = m;
      while (mm) {
        const int k = __ffs(mm) - 1;
        mm &= mm - 1;
        const int bin = (words[k >> 2] >> ((k & 3) * 8)) & 0xFF;
        const int cell = 2 * (k * n_bins + bin);
        atomicAdd(lg + cell, (double)v.x);
        atomicAdd(lp + cell, hq);
      }
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < tot; idx += blockDim.x) {
    const double g = lg[2 * idx];
    const unsigned long long pk = lp[2 * idx];
    if (pk == 0ull && g == 0.0) continue;
    const int k = idx / n_bins;
    const int f = fg * 32 + k;
    if (f >= F) continue;
    const int bin = idx - k * n_bins;
    float* p = hist + ((int64_t)slot_abs * F + f) * (n_bins * 3)
               + bin * 3;
    atomicAdd(p, (float)g);
    atomicAdd(p + 1, (float)((double)(pk & kHMask) * (double)kHInvScale));
    atomicAdd(p + 2, (float)(pk >> 44));
  }
}

// ---------------------------------------------------------------------------
// Masked histogram zeroing: with per-slot feature sampling only the
// sampled (slot, feature) cells (plus feature 0, the totals source) are
// ever written or read, so zeroing the whole [slots, F, 256, 3] buffer
// (measured ~10% of the RF step) is wasted — clear just the live cells.
// ---------------------------------------------------------------------------
__global__ void zero_hist_masked_kernel(float* __restrict__ hist,
                                        const uint16_t* __restrict__
                                            maskbits,
                                        int F, int n_bins, int ns) {
  const int fg = blockIdx.x;
  const int slot = blockIdx.z;
  const int F16 = (F + 15) / 16;
  unsigned m = maskbits[(int64_t)slot * F16 + fg];
  while (m) {
    const int k = __ffs(m) - 1;
    m &= m - 1;
    const int f = fg * 16 + k;
    if (f >= F) continue;
    float* p = hist + ((int64_t)slot * F + f) * (n_bins * 3);
    for (int i = threadIdx.x; i < n_bins * 3; i += blockDim.x) p[i] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// Feature-interleaved gathered histograms: bins16 stores the binned
// matrix as [ceil(F/16)][N][16] u8, so ONE 16-byte load fetches a row's
// bins for 16 features (vs 16 scattered single-byte gathers from the
// row-major [F][N] layout — the measured bottleneck of the deep-level
// partitioned sweep). Each block owns (16 features) x (one slot GROUP of
// `spg` slots, blockIdx.z) over a contiguous row range of `row_order`;
// LDS holds [16][spg][256] packed {f64 g, u64 h|cnt} cells
// (spg=2 -> 128 KiB of the 160 KiB LDS).
// ---------------------------------------------------------------------------
__global__ void hist_build_gathered16_kernel(
    const uint8_t* __restrict__ bins16, const float2* __restrict__ gh,
    const int32_t* __restrict__ node_ids,
    const int32_t* __restrict__ slot_map,
    const int32_t* __restrict__ row_order,
    const int64_t* __restrict__ group_offs, float* __restrict__ hist,
    const uint16_t* __restrict__ maskbits,  // [n_slots][F16] or null
    int64_t N, int F, int n_bins, int level_base, int level_size,
    int win0, int spg, int n_chunks) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* lg = reinterpret_cast<double*>(smem);
  unsigned long long* lp =
      reinterpret_cast<unsigned long long*>(smem) + 1;
  const int fg = blockIdx.x;
  const int z = blockIdx.z;
  const int tot = 16 * spg * n_bins;
  {
    unsigned long long* zz = reinterpret_cast<unsigned long long*>(smem);
    for (int i = threadIdx.x; i < tot * 2; i += blockDim.x) zz[i] = 0ull;
  }
  __syncthreads();
  const int64_t r0 = group_offs[z];
  const int64_t r1 = group_offs[z + 1];
  const int64_t per = (r1 - r0 + n_chunks - 1) / n_chunks;
  const int64_t j0 = r0 + (int64_t)blockIdx.y * per;
  const int64_t j1 = min(j0 + per, r1);
  const uint4* fb =
      reinterpret_cast<const uint4*>(bins16 + (int64_t)fg * N * 16);
  const int slot_lo = z * spg;
  const int F16 = (F + 15) / 16;
  unsigned m_of_slot[8];  // spg <= 8
  if (maskbits != nullptr) {
    for (int sI = 0; sI < spg; ++sI)
      m_of_slot[sI] =
          maskbits[(int64_t)(slot_lo + sI) * F16 + fg];
  }
  for (int64_t j = j0 + threadIdx.x; j < j1; j += blockDim.x) {
    const int row = row_order[j];
    const int rel = node_ids[row] - level_base;
    if (rel < 0 || rel >= level_size) continue;
    const int slot = slot_map[rel] - win0 - slot_lo;
    if (slot < 0 || slot >= spg) continue;
    unsigned m = 0xFFFFu;
    if (maskbits != nullptr) {
      m = m_of_slot[slot];
      if (m == 0u) continue;  // no sampled feature in this group
    }
    const float2 v = gh[row];
    const unsigned long long hq =
        (unsigned long long)(v.y * kHScale + 0.5f) |
        ((unsigned long long)(v.y != 0.f) << 44);
    const uint4 b = fb[row];
    const unsigned words[4] = {b.x, b.y, b.z, b.w};
    while (m) {
      const int k = __ffs(m) - 1;
      m &= m - 1;
      const int bin = (words[k >> 2] >> ((k & 3) * 8)) & 0xFF;
      const int cell = 2 * ((k * spg + slot) * n_bins + bin);
      atomicAdd(lg + cell, (double)v.x);
      atomicAdd(lp + cell, hq);
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < tot; idx += blockDim.x) {
    const double g = lg[2 * idx];
    const unsigned long long pk = lp[2 * idx];
    if (pk == 0ull && g == 0.0) continue;
    const int k = idx / (spg * n_bins);
    const int f = fg * 16 + k;
    if (f >= F) continue;
    const int rem = idx - k * spg * n_bins;
    const int slot = rem / n_bins;
    const int bin = rem - slot * n_bins;
    float* p = hist + ((int64_t)(slot_lo + slot) * F + f) * (n_bins * 3)
               + bin * 3;
    atomicAdd(p, (float)g);
    atomicAdd(p + 1, (float)((double)(pk & kHMask) * (double)kHInvScale));
    atomicAdd(p + 2, (float)(pk >> 44));
  }
}

// ---------------------------------------------------------------------------
// Split scan, stage A: one block per (slot, feature). Each thread owns one
// bin; inclusive prefix sums of {g,h,c} over bins via LDS Hillis-Steele
// (deterministic), then per-boundary gain and a deterministic block argmax.
// Writes the per-(slot,feature) best {gain, bin}, and (f==0) node totals
// into node_stats[abs_node]. Gain (2nd-order, reference use_hessian_gain
// semantics): gain = GL^2/(HL+l2) + GR^2/(HR+l2) - G^2/(H+l2).
// ---------------------------------------------------------------------------
// Bitonic sort of the n_bins (key, idx) pairs in LDS; ascending by
// (key, idx) — deterministic under ties. Used for categorical features:
// categories ordered by gradient statistic G/(H + smooth), then scanned
// like a numerical feature (the reference's CART one-vs-rest ordering,
// splitter_scanner.h:859 sorting variants; same algorithm as LightGBM's
// categorical handling). Empty bins carry key=+inf (sort last).
__device__ inline void bitonic_sort_bins(float* key, short* idx, int n,
                                         int tid) {
  for (int k = 2; k <= n; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      const int ixj = tid ^ j;
      if (ixj > tid) {
        const bool up = (tid & k) == 0;
        const float a = key[tid], bval = key[ixj];
        const short ia = idx[tid], ib = idx[ixj];
        const bool swap = up ? (a > bval || (a == bval && ia > ib))
                             : (a < bval || (a == bval && ia < ib));
        if (swap) {
          key[tid] = bval; key[ixj] = a;
          idx[tid] = ib; idx[ixj] = ia;
        }
      }
      __syncthreads();
    }
  }
}

__global__ void split_scan_feat_kernel(const float* __restrict__ hist,
                                       const int32_t* __restrict__ abs_of_slot,
                                       float* __restrict__ node_stats,
                                       float* __restrict__ best_gain_nf,
                                       int32_t* __restrict__ best_bin_nf,
                                       const uint8_t* __restrict__ feat_mask,
                                       const uint8_t* __restrict__ cat_flags,
                                       const int8_t* __restrict__ mono,
                                       int32_t* __restrict__ na_meanb_nf,
                                       int F, int n_bins, int slot0,
                                       SplitParams sp) {
  const int slot = blockIdx.x;
  const int f = blockIdx.y;
  const int b = threadIdx.x;  // blockDim.x == n_bins (power of two <= 256)
  // masked-out (slot, feature) blocks exit before reading the histogram
  // (f == 0 always scans: it computes the node totals) — with RF's
  // sqrt(F) per-node sampling this skips ~95% of the scan traffic
  if (f != 0 && feat_mask != nullptr &&
      !feat_mask[(int64_t)(slot0 + slot) * F + f]) {
    if (b == 0) {
      best_gain_nf[(int64_t)slot * F + f] = -1e30f;
      best_bin_nf[(int64_t)slot * F + f] = 0;
    }
    return;
  }
  __shared__ float sg[kMaxBins], sh[kMaxBins], sc[kMaxBins];
  const float* hp = hist + ((int64_t)slot * F + f) * (n_bins * 3);
  float rg0 = hp[b * 3];
  float rh0 = hp[b * 3 + 1];
  float rc0 = hp[b * 3 + 2];
  if (cat_flags != nullptr && cat_flags[f]) {
    // categorical: sort bins by G/(H+smooth) before scanning
    __shared__ float skey[kMaxBins];
    __shared__ short sidx[kMaxBins];
    skey[b] = (rc0 > 0.f) ? rg0 / (rh0 + sp.cat_smooth) : 1e30f;
    sidx[b] = (short)b;
    sg[b] = rg0; sh[b] = rh0; sc[b] = rc0;
    __syncthreads();
    bitonic_sort_bins(skey, sidx, n_bins, b);
    const int src = sidx[b];
    rg0 = sg[src]; rh0 = sh[src]; rc0 = sc[src];
    __syncthreads();
  }
  sg[b] = rg0;
  sh[b] = rh0;
  sc[b] = rc0;
  __syncthreads();
  if (sp.na_mode && (cat_flags == nullptr || !cat_flags[f])) {
    // LOCAL_IMPUTATION (decision_tree.proto:85-103): bin 255 holds the
    // node's MISSING rows; fold them into the node-local mean bin and
    // remember it so the winning split can store the na direction
    __shared__ int s_meanb;
    if (b == 0) {
      float wsum = 0.f, csum = 0.f;
      for (int i = 0; i < n_bins - 1; ++i) {
        wsum += (float)i * sc[i];
        csum += sc[i];
      }
      s_meanb = csum > 0.f ? (int)(wsum / csum + 0.5f) : 0;
      if (s_meanb > n_bins - 2) s_meanb = n_bins - 2;
      if (na_meanb_nf != nullptr)
        na_meanb_nf[(int64_t)slot * F + f] = s_meanb;
    }
    __syncthreads();
    if (b == s_meanb) {
      sg[b] += sg[n_bins - 1];
      sh[b] += sh[n_bins - 1];
      sc[b] += sc[n_bins - 1];
    }
    __syncthreads();
    if (b == n_bins - 1) {
      sg[b] = 0.f;
      sh[b] = 0.f;
      sc[b] = 0.f;
    }
    __syncthreads();
  }
  for (int off = 1; off < n_bins; off <<= 1) {
    float tg = 0.f, th = 0.f, tc = 0.f;
    if (b >= off) { tg = sg[b - off]; th = sh[b - off]; tc = sc[b - off]; }
    __syncthreads();
    sg[b] += tg; sh[b] += th; sc[b] += tc;
    __syncthreads();
  }
  const float G = sg[n_bins - 1], H = sh[n_bins - 1], C = sc[n_bins - 1];
  // C > 0 guard: empty slots (dense-mode nodes that were pruned) must not
  // zero out node stats already written by their parent's split.
  if (f == 0 && b == 0 && C > 0.f) {
    float* ns = node_stats + (int64_t)abs_of_slot[slot0 + slot] * 3;
    ns[0] = G; ns[1] = H; ns[2] = C;
  }
  // Candidate split: left = bins [0..b], valid for b in [0, n_bins-2].
  // feat_mask (per-slot feature sampling, reference num_candidate_attributes)
  // disables the whole feature.
  float gain = -1e30f;
  if (b < n_bins - 1 &&
      (feat_mask == nullptr || feat_mask[(slot0 + slot) * F + f])) {
    const float GL = sg[b], HL = sh[b], CL = sc[b];
    const float GR = G - GL, HR = H - HL, CR = C - CL;
    if (CL >= sp.min_examples && CR >= sp.min_examples &&
        HL >= sp.min_hessian && HR >= sp.min_hessian) {
      bool ok = true;
      if (mono != nullptr && mono[f] != 0) {
        // monotonic constraint (reference monotonic_constraints;
        // XGBoost-style): reject splits whose child values violate the
        // declared direction
        const float wl = -l1_thresh(GL, sp.lambda_l1) / (HL + sp.lambda_l2);
        const float wr = -l1_thresh(GR, sp.lambda_l1) / (HR + sp.lambda_l2);
        ok = (mono[f] > 0) ? (wl <= wr) : (wl >= wr);
      }
      if (ok) {
        const float tl = l1_thresh(GL, sp.lambda_l1);
        const float tr = l1_thresh(GR, sp.lambda_l1);
        const float tp = l1_thresh(G, sp.lambda_l1);
        gain = tl * tl / (HL + sp.lambda_l2) + tr * tr / (HR + sp.lambda_l2) -
               tp * tp / (H + sp.lambda_l2);
      }
    }
  }
  // Deterministic argmax reduce: higher gain wins; ties -> smaller bin.
  __shared__ float rg[kMaxBins];
  __shared__ int rb[kMaxBins];
  rg[b] = gain; rb[b] = b;
  __syncthreads();
  for (int off = n_bins >> 1; off > 0; off >>= 1) {
    if (b < off) {
      const float og = rg[b + off];
      const int ob = rb[b + off];
      if (og > rg[b] || (og == rg[b] && ob < rb[b])) { rg[b] = og; rb[b] = ob; }
    }
    __syncthreads();
  }
  if (b == 0) {
    best_gain_nf[(int64_t)slot * F + f] = rg[0];
    best_bin_nf[(int64_t)slot * F + f] = rb[0];
  }
}

// ---------------------------------------------------------------------------
// Split scan, stage B: one block per slot. Deterministic argmax over
// features (ties -> smaller feature), then recompute left-child stats for
// the winner and emit the split + child node stats. best_feat = -1 => leaf.
// ---------------------------------------------------------------------------
__global__ void split_select_kernel(const float* __restrict__ hist,
                                    const int32_t* __restrict__ abs_of_slot,
                                    const float* __restrict__ best_gain_nf,
                                    const int32_t* __restrict__ best_bin_nf,
                                    float* __restrict__ node_stats,
                                    int32_t* __restrict__ best_feat,
                                    int32_t* __restrict__ best_bin,
                                    float* __restrict__ best_gain,
                                    const uint8_t* __restrict__ cat_flags,
                                    unsigned long long* __restrict__ masks,
                                    const int8_t* __restrict__ mono,
                                    float* __restrict__ node_bounds,
                                    const int32_t* __restrict__ na_meanb_nf,
                                    uint8_t* __restrict__ tree_na,
                                    int F, int n_bins, int slot0,
                                    SplitParams sp) {
  const int slot = blockIdx.x;
  const int t = threadIdx.x;
  __shared__ float rg[kBlock];
  __shared__ int rf[kBlock];
  float g = -1e30f;
  int bf = -1;
  for (int f = t; f < F; f += blockDim.x) {
    const float fg = best_gain_nf[(int64_t)slot * F + f];
    if (fg > g) { g = fg; bf = f; }
  }
  rg[t] = g; rf[t] = bf;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (t < off) {
      const float og = rg[t + off];
      const int of = rf[t + off];
      if (og > rg[t] || (og == rg[t] && of >= 0 && (rf[t] < 0 || of < rf[t]))) {
        rg[t] = og; rf[t] = of;
      }
    }
    __syncthreads();
  }
  __shared__ int w_f, w_bin, w_abs;
  if (t == 0) {
    const float gain = rg[0];
    const int f = rf[0];
    const int out = slot0 + slot;
    if (f < 0 || gain <= sp.min_gain) {
      best_feat[out] = -1;
      best_bin[out] = 0;
      best_gain[out] = 0.f;
      w_f = -1;
    } else {
      const int bin = best_bin_nf[(int64_t)slot * F + f];
      best_feat[out] = f;
      best_bin[out] = bin;
      best_gain[out] = gain;
      w_f = f;
      w_bin = bin;
      w_abs = abs_of_slot[out];
    }
  }
  __syncthreads();
  const int f = w_f;
  if (f < 0) return;
  const bool is_cat = cat_flags != nullptr && cat_flags[f];
  const float* hp = hist + ((int64_t)slot * F + f) * (n_bins * 3);
  const int abs_node = w_abs;
  if (!is_cat) {
    const int bin = w_bin;
    // Small grids (shallow GBT levels: few slots = few blocks) are
    // latency-bound on the serial thread-0 prefix — parallelize it
    // across the block (~2/3 of the kernel's 14.5 us there). Large
    // grids (deep RF levels: 1000+ blocks) already hide that latency
    // with block parallelism and only pay the reduction's syncs, so
    // they keep the serial form.
    float GL, HL, CL;
    if (gridDim.x <= 64) {
      float part[3] = {0.f, 0.f, 0.f};
      for (int bb = t; bb <= bin; bb += blockDim.x) {
        part[0] += hp[bb * 3];
        part[1] += hp[bb * 3 + 1];
        part[2] += hp[bb * 3 + 2];
      }
      float red[3];
      for (int c = 0; c < 3; ++c) {
        rg[t] = part[c];
        __syncthreads();
        for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
          if (t < off) rg[t] += rg[t + off];
          __syncthreads();
        }
        red[c] = rg[0];
        __syncthreads();
      }
      if (t != 0) return;
      GL = red[0];
      HL = red[1];
      CL = red[2];
    } else {
      if (t != 0) return;
      GL = HL = CL = 0.f;
      for (int bb = 0; bb <= bin; ++bb) {
        GL += hp[bb * 3];
        HL += hp[bb * 3 + 1];
        CL += hp[bb * 3 + 2];
      }
    }
    if (sp.na_mode && na_meanb_nf != nullptr) {
      // the scan folded the NA bin (255) into the node-local mean bin;
      // mirror that here and record the na direction for routing
      const int meanb = na_meanb_nf[(int64_t)slot * F + f];
      if (tree_na != nullptr)
        tree_na[abs_node] = meanb > bin ? (uint8_t)1 : (uint8_t)0;
      if (meanb <= bin) {
        GL += hp[(n_bins - 1) * 3];
        HL += hp[(n_bins - 1) * 3 + 1];
        CL += hp[(n_bins - 1) * 3 + 2];
      }
    }
    const float* ns = node_stats + (int64_t)abs_node * 3;
    float* nl = node_stats + (int64_t)(2 * abs_node + 1) * 3;
    float* nr = node_stats + (int64_t)(2 * abs_node + 2) * 3;
    nl[0] = GL; nl[1] = HL; nl[2] = CL;
    nr[0] = ns[0] - GL; nr[1] = ns[1] - HL; nr[2] = ns[2] - CL;
    if (node_bounds != nullptr) {
      const float lo = node_bounds[2 * abs_node];
      const float hi = node_bounds[2 * abs_node + 1];
      float llo = lo, lhi = hi, rlo = lo, rhi = hi;
      if (mono != nullptr && mono[f] != 0) {
        const float wl =
            fminf(fmaxf(-l1_thresh(GL, sp.lambda_l1)
                        / (HL + sp.lambda_l2), lo), hi);
        const float wr = fminf(
            fmaxf(-l1_thresh(ns[0] - GL, sp.lambda_l1)
                  / (ns[1] - HL + sp.lambda_l2), lo), hi);
        const float mid = 0.5f * (wl + wr);
        if (mono[f] > 0) { lhi = mid; rlo = mid; }
        else { llo = mid; rhi = mid; }
      }
      node_bounds[2 * (2 * abs_node + 1)] = llo;
      node_bounds[2 * (2 * abs_node + 1) + 1] = lhi;
      node_bounds[2 * (2 * abs_node + 2)] = rlo;
      node_bounds[2 * (2 * abs_node + 2) + 1] = rhi;
    }
    return;
  }
  // Categorical winner: rebuild the sorted order (same deterministic sort
  // as stage A), emit the "goes RIGHT" category bitmask = sorted ranks >
  // best rank with count > 0 (unseen categories default LEFT, matching
  // the reference's negative-child policy for absent values), and the
  // child stats from the sorted prefix.
  __shared__ float skey[kMaxBins];
  __shared__ short sidx[kMaxBins];
  __shared__ float sg[kMaxBins], sh[kMaxBins], sc[kMaxBins];
  const int b = t;  // blockDim == kBlock == kMaxBins == n_bins required
  const float bg = hp[b * 3], bh = hp[b * 3 + 1], bc = hp[b * 3 + 2];
  skey[b] = (bc > 0.f) ? bg / (bh + sp.cat_smooth) : 1e30f;
  sidx[b] = (short)b;
  sg[b] = bg; sh[b] = bh; sc[b] = bc;
  __syncthreads();
  bitonic_sort_bins(skey, sidx, n_bins, b);
  const int src = sidx[b];
  const float mg = sg[src], mh = sh[src], mc = sc[src];
  __syncthreads();
  // mask: ranks > w_bin with examples go right
  __shared__ unsigned long long lmask[kMaxBins / 64];
  if (b < n_bins / 64) lmask[b] = 0ull;
  __syncthreads();
  if (b > w_bin && mc > 0.f)
    atomicOr(&lmask[src >> 6], 1ull << (src & 63));
  // left-child stats: inclusive prefix over sorted ranks <= w_bin
  sg[b] = (b <= w_bin) ? mg : 0.f;
  sh[b] = (b <= w_bin) ? mh : 0.f;
  sc[b] = (b <= w_bin) ? mc : 0.f;
  __syncthreads();
  for (int off = n_bins >> 1; off > 0; off >>= 1) {
    if (b < off) {
      sg[b] += sg[b + off]; sh[b] += sh[b + off]; sc[b] += sc[b + off];
    }
    __syncthreads();
  }
  if (b < n_bins / 64 && masks != nullptr)
    masks[(int64_t)abs_node * (kMaxBins / 64) + b] = lmask[b];
  if (b == 0) {
    const float GL = sg[0], HL = sh[0], CL = sc[0];
    const float* ns = node_stats + (int64_t)abs_node * 3;
    float* nl = node_stats + (int64_t)(2 * abs_node + 1) * 3;
    float* nr = node_stats + (int64_t)(2 * abs_node + 2) * 3;
    nl[0] = GL; nl[1] = HL; nl[2] = CL;
    nr[0] = ns[0] - GL; nr[1] = ns[1] - HL; nr[2] = ns[2] - CL;
    if (node_bounds != nullptr) {  // categorical: children inherit bounds
      node_bounds[2 * (2 * abs_node + 1)] = node_bounds[2 * abs_node];
      node_bounds[2 * (2 * abs_node + 1) + 1] =
          node_bounds[2 * abs_node + 1];
      node_bounds[2 * (2 * abs_node + 2)] = node_bounds[2 * abs_node];
      node_bounds[2 * (2 * abs_node + 2) + 1] =
          node_bounds[2 * abs_node + 1];
    }
  }
}

// ---------------------------------------------------------------------------
// Dense-mode device planning: for every node of a level (slot == rel), decide
// build/derive/skip from the previous level's splits and the children stats
// already on device — no host round-trip per level. derived: sibling's
// histogram will be parent - computed sibling (histogram-subtraction trick).
//   build_map[rel] = rel if histogram must be built else -1
//   derived[rel]   = 1 if hist[rel] = hist_prev[rel>>1] - hist[rel^1]
// ---------------------------------------------------------------------------
__global__ void plan_level_kernel(const float* __restrict__ node_stats,
                                  const int32_t* __restrict__ prev_best_feat,
                                  int level_base, int level_size, int need,
                                  int use_sub, int32_t* __restrict__ build_map,
                                  uint8_t* __restrict__ derived) {
  const int rel = blockIdx.x * blockDim.x + threadIdx.x;
  if (rel >= level_size) return;
  const int a = level_base + rel;
  const int split = prev_best_feat[rel >> 1] >= 0;
  const float ca = node_stats[(int64_t)a * 3 + 2];
  const int sib_rel = rel ^ 1;
  const float cs = node_stats[(int64_t)(level_base + sib_rel) * 3 + 2];
  const int elig_a = split && ca >= (float)need;
  const int elig_s = split && cs >= (float)need;
  // "a" is a LEFT child iff abs index is odd
  const int is_right = (a & 1) == 0;
  const int dv = use_sub && elig_a && elig_s &&
                 (ca > cs || (ca == cs && is_right));
  derived[rel] = (uint8_t)dv;
  build_map[rel] = (elig_a && !dv) ? rel : -1;
}

// hist[r] = hist_prev[r>>1] - hist[r^1] for derived slots (dense mode).
__global__ void subtract_hist_kernel(float* __restrict__ hist,
                                     const float* __restrict__ hist_prev,
                                     const uint8_t* __restrict__ derived,
                                     int F, int n_bins) {
  const int rel = blockIdx.x;
  if (!derived[rel]) return;
  const int64_t cells = (int64_t)F * n_bins * 3;
  float* dst = hist + (int64_t)rel * cells;
  const float* par = hist_prev + (int64_t)(rel >> 1) * cells;
  const float* sib = hist + (int64_t)(rel ^ 1) * cells;
  for (int64_t k = (int64_t)blockIdx.y * blockDim.x + threadIdx.x; k < cells;
       k += (int64_t)gridDim.y * blockDim.x)
    dst[k] = par[k] - sib[k];
}

// ---------------------------------------------------------------------------
// Route examples through this level's chosen splits: bin > split_bin ->
// right child. Examples at closed/leaf nodes keep their node id ("parked").
// best_feat/best_bin are indexed by SLOT (whole level, all chunks).
// ---------------------------------------------------------------------------
__global__ void update_node_ids_kernel(const uint8_t* __restrict__ bins,
                                       int32_t* __restrict__ node_ids,
                                       const int32_t* __restrict__ slot_map,
                                       const int32_t* __restrict__ best_feat,
                                       const int32_t* __restrict__ best_bin,
                                       const uint8_t* __restrict__ cat_flags,
                                       const unsigned long long* __restrict__
                                           masks,
                                       const uint8_t* __restrict__ tree_na,
                                       int64_t N, int level_base,
                                       int level_size) {
  // 4-row unroll: four independent (node -> split -> bin) load chains in
  // flight per lane instead of one serialized chain.
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t k = tid;
  const int64_t bulk_end = N - 3 * stride;
  for (; k < bulk_end; k += 4 * stride) {
    int nid[4], f[4], sbin[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) nid[u] = node_ids[k + u * stride];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int rel = nid[u] - level_base;
      f[u] = -1;
      if (rel >= 0 && rel < level_size) {
        const int slot = slot_map[rel];
        if (slot >= 0) {
          f[u] = best_feat[slot];
          sbin[u] = best_bin[slot];
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (f[u] < 0) continue;
      const int b = bins[(int64_t)f[u] * N + k + u * stride];
      int right;
      if (cat_flags != nullptr && cat_flags[f[u]])
        right = (int)((masks[(int64_t)nid[u] * (kMaxBins / 64) + (b >> 6)]
                       >> (b & 63)) & 1ull);
      else if (tree_na != nullptr && b == kMaxBins - 1)
        right = tree_na[nid[u]];  // missing row: stored na direction
      else
        right = b > sbin[u];
      node_ids[k + u * stride] = 2 * nid[u] + 1 + right;
    }
  }
  for (; k < N; k += stride) {
    const int nid = node_ids[k];
    const int rel = nid - level_base;
    if (rel < 0 || rel >= level_size) continue;
    const int slot = slot_map[rel];
    if (slot < 0) continue;
    const int f = best_feat[slot];
    if (f < 0) continue;  // leaf: park
    const int b = bins[(int64_t)f * N + k];
    int right;
    if (cat_flags != nullptr && cat_flags[f])
      right = (int)((masks[(int64_t)nid * (kMaxBins / 64) + (b >> 6)]
                     >> (b & 63)) & 1ull);
    else if (tree_na != nullptr && b == kMaxBins - 1)
      right = tree_na[nid];
    else
      right = b > best_bin[slot];
    node_ids[k] = 2 * nid + 1 + right;
  }
}

// leaf value for every materialized node: -G / (H + l2). Harmless for
// internal nodes (examples only ever point at leaves).
__global__ void leaf_values_kernel(const float* __restrict__ node_stats,
                                   const float* __restrict__ node_bounds,
                                   float* __restrict__ leaf_values,
                                   int total_nodes, float lambda_l2,
                                   float lambda_l1) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total_nodes) return;
  const float* ns = node_stats + (int64_t)i * 3;
  float v = (ns[1] != 0.f)
                ? (-l1_thresh(ns[0], lambda_l1) / (ns[1] + lambda_l2))
                : 0.f;
  if (node_bounds != nullptr)
    v = fminf(fmaxf(v, node_bounds[2 * i]), node_bounds[2 * i + 1]);
  leaf_values[i] = v;
}

__global__ void update_preds_kernel(float* __restrict__ preds,
                                    const int32_t* __restrict__ node_ids,
                                    const float* __restrict__ leaf_values,
                                    int64_t N, float shrinkage) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    const int nid = node_ids[k];
    if (nid >= 0) preds[k] += shrinkage * leaf_values[nid];
  }
}

// Binary log-loss + accuracy partial sums (validation/early stopping),
// block-reduced then atomically merged into out[0]=loss_sum, out[1]=correct.
__global__ void binary_logloss_kernel(const float* __restrict__ preds,
                                      const float* __restrict__ labels,
                                      float* __restrict__ out, int64_t N) {
  __shared__ float sl[kBlock], sa[kBlock];
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float loss = 0.f, acc = 0.f;
  for (int64_t k = i; k < N; k += stride) {
    const float m = preds[k];
    const float y = labels[k];
    const float z = y > 0.5f ? -m : m;
    loss += (z > 0.f) ? z + __logf(1.f + __expf(-z)) : __logf(1.f + __expf(z));
    acc += ((m > 0.f) == (y > 0.5f)) ? 1.f : 0.f;
  }
  sl[threadIdx.x] = loss; sa[threadIdx.x] = acc;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      sl[threadIdx.x] += sl[threadIdx.x + off];
      sa[threadIdx.x] += sa[threadIdx.x + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicAdd(&out[0], sl[0]);
    atomicAdd(&out[1], sa[0]);
  }
}

// ---------------------------------------------------------------------------
// Host launchers (raw pointers + explicit stream; exported via pybind).
// ---------------------------------------------------------------------------
static inline int row_chunks(int64_t N, int F, int max_blocks = 8192) {
  // Enough blocks to fill 256 CUs x several waves, but bounded: every
  // block merges (and first zeroes) its LDS histogram, so block count is
  // also merge traffic. Measured on MI355X (profiles/
  // kernel_stats_r02_final.md sweep): at small shards the kernel is
  // merge-bound, and ~64k rows/chunk is the sweet spot at every shard
  // size of the 8-GPU strong-scaling ladder (1.375M: 73->21 chunks =
  // +16% trees/s; 11M unchanged because the max_blocks cap binds first).
  int per_f = (int)((max_blocks + F - 1) / F);
  int64_t target = (N + 65535) >> 16;           // >=64k rows per chunk
  const int64_t fill = (512 + F - 1) / F;       // ...but keep >=512 WGs
  if (target < fill) target = fill;
  const int64_t max_chunks = (N + 1023) >> 10;  // never <1k rows/chunk
  int chunks = (int)(target < per_f ? target : per_f);
  if (chunks > max_chunks) chunks = (int)max_chunks;
  if (chunks < 1) chunks = 1;
  // count/h-field packing requires rows_per_block <= 2^19
  const int min_chunks = (int)((N + (1 << 19) - 1) >> 19);
  if (chunks < min_chunks) chunks = min_chunks;
  return chunks;
}

static inline int hist_block_threads() {
  static int cached = 0;
  if (cached == 0) {
    const char* e = getenv("YDFA_HIST_BLOCK");
    int v = e ? atoi(e) : 0;
    // 1024 threads/block measured fastest on MI355X (fewer blocks => less
    // merge traffic; 16 waves/CU still hide the streaming latency)
    if (v != 256 && v != 512 && v != 1024) v = 1024;
    cached = v;
  }
  return cached;
}

extern "C" {

void gpu_bin_data(const float* x, const float* boundaries, uint8_t* out,
                  int64_t N, int F, int n_cuts, int na_to_255,
                  void* stream) {
  const int chunks = row_chunks(N, F);
  const int64_t rpb = (N + chunks - 1) / chunks;
  hipLaunchKernelGGL(bin_data_kernel, dim3(F, chunks), dim3(kBlock), 0,
                     (hipStream_t)stream, x, boundaries, out, N, F, n_cuts,
                     rpb, na_to_255);
}

static int elem_grid(int64_t N, int cap = 2048) {
  int grid = (int)((N + kBlock - 1) / kBlock);
  if (grid > cap) grid = cap;
  if (grid < 1) grid = 1;
  return grid;
}

void gpu_grad_hess(const float* preds, const float* labels, float* gh,
                   int64_t N, int loss, void* stream) {
  hipLaunchKernelGGL(grad_hess_kernel, dim3(elem_grid(N)), dim3(kBlock), 0,
                     (hipStream_t)stream, preds, labels, (float2*)gh, N,
                     loss);
}

void gpu_grad_hess_softmax(const float* preds, const float* labels, float* gh,
                           int64_t N, int n_classes, int cls, void* stream) {
  hipLaunchKernelGGL(grad_hess_softmax_kernel, dim3(elem_grid(N)),
                     dim3(kBlock), 0, (hipStream_t)stream, preds, labels,
                     (float2*)gh, N, n_classes, cls);
}

void gpu_weighted_target(const float* labels, const float* weights, float* gh,
                         int64_t N, void* stream) {
  hipLaunchKernelGGL(weighted_target_kernel, dim3(elem_grid(N)), dim3(kBlock),
                     0, (hipStream_t)stream, labels, weights, (float2*)gh, N);
}

// hist must be zeroed by the caller. Slices [slot0, slot0+n_slots) into
// LDS-sized groups internally; hist is the base pointer for slot0.
void gpu_hist_build(const uint8_t* bins, const float* gh,
                    const int32_t* node_ids, const int32_t* slot_map,
                    float* hist, int64_t N, int F, int n_bins, int level_base,
                    int level_size, int slot0, int n_slots,
                    int filtered_hint, uint8_t* grp_scratch, void* stream) {
  // Stage the level's slot map in LDS when it fits comfortably (removes a
  // dependent global load per row visit); cap at 32 KiB so the histogram
  // region keeps >= ~32 slots at 16 B/bin.
  const size_t map_bytes_full = (size_t)level_size * sizeof(int32_t);
  const int lds_map = map_bytes_full <= 32 * 1024 ? 1 : 0;
  const size_t budget = 160 * 1024 - (lds_map ? map_bytes_full : 0);
  int max_lds_slots = (int)(budget / ((size_t)n_bins * 16));
  if (max_lds_slots < 1) max_lds_slots = 1;
  // Optional cap: smaller slot groups shrink per-block LDS (e.g. a
  // 32-slot level at 16 B/bin needs 131 KB -> 1 WG/CU; capping at 16
  // gives 64 KB -> 2 WG/CU at the cost of a second filtered pass).
  static int max_slots_env = -1;
  if (max_slots_env < 0) {
    const char* e = getenv("YDFA_HIST_MAX_SLOTS");
    max_slots_env = e ? atoi(e) : 0;
  }
  if (max_slots_env > 0 && max_lds_slots > max_slots_env)
    max_lds_slots = max_slots_env;
  const int group = n_slots < max_lds_slots ? n_slots : max_lds_slots;
  // multi-feature blocks: when the whole level fits with room to spare,
  // each block histograms fpb features from ONE pass over the rows
  static int max_fpb = -1;
  if (max_fpb < 0) {
    const char* e = getenv("YDFA_HIST_FPB");
    max_fpb = e ? atoi(e) : 1;
    if (max_fpb < 1) max_fpb = 1;
  }
  int fpb = 1;
  if (group == n_slots) {
    fpb = (int)(budget / ((size_t)n_slots * n_bins * 16));
    if (fpb > max_fpb) fpb = max_fpb;
    if (fpb > F) fpb = F;
    if (fpb < 1) fpb = 1;
  }
  const int n_fgroups = (F + fpb - 1) / fpb;
  const int threads = hist_block_threads();
  static int max_blocks_env = -1;
  if (max_blocks_env < 0) {
    const char* e = getenv("YDFA_HIST_MAX_BLOCKS");
    max_blocks_env = e ? atoi(e) : 0;
  }
  int chunks;
  if (max_blocks_env > 0) {
    // explicit override (sweeps): block-cap-driven chunking with the
    // legacy 1k-row floor, bypassing the 64k-rows-per-chunk target
    const int per_f = (max_blocks_env + n_fgroups - 1) / n_fgroups;
    const int64_t mx = (N + 1023) >> 10;
    chunks = (int)(mx < per_f ? mx : per_f);
    const int min_c = (int)((N + (1 << 19) - 1) >> 19);
    if (chunks < min_c) chunks = min_c;
    if (chunks < 1) chunks = 1;
  } else {
    chunks = row_chunks(N, n_fgroups, 8192 * 256 / threads);
  }
  {
    static int64_t max_rpb = -1;
    if (max_rpb < 0) {
      const char* e = getenv("YDFA_HIST_CHUNK_KB");  // gh bytes per chunk
      max_rpb = (e ? atoi(e) : 3072) * 1024 / 8;
      if (max_rpb <= 0) max_rpb = 1 << 30;
    }
    const int min_chunks = (int)((N + max_rpb - 1) / max_rpb);
    if (chunks < min_chunks) chunks = min_chunks;
  }
  static int swizzle = -1;
  if (swizzle < 0) {
    const char* e = getenv("YDFA_HIST_SWIZZLE");
    swizzle = e ? atoi(e) : 1;
  }
  const int64_t rpb = (N + chunks - 1) / chunks;
  // the u8 prefilter is only valid when every launch covers whole groups
  // (slot0 must be group-aligned). Measured SLOWER than the node-id path
  // on the 1Mx500 depth-16 RF config (deep levels are issue-bound, not
  // bytes-bound), so it is opt-in via YDFA_HIST_GRP=1.
  static int grp_enabled = -1;
  if (grp_enabled < 0) {
    const char* e = getenv("YDFA_HIST_GRP");
    grp_enabled = e ? atoi(e) : 0;
  }
  const int use_grp = (grp_enabled && grp_scratch != nullptr &&
                       n_slots > group && group <= 255 &&
                       (slot0 % group) == 0) ? 1 : 0;
  if (use_grp) {
    hipLaunchKernelGGL(row_group_kernel, dim3(elem_grid(N, 4096)),
                       dim3(kBlock), 0, (hipStream_t)stream, node_ids,
                       slot_map, grp_scratch, N, level_base, level_size,
                       group);
  }
  // slot8 mode (single-launch levels where the caller hints that many
  // rows are not built, i.e. histogram-subtraction levels): precompute a
  // per-row u8 slot id so every feature-group block streams 1 B/row for
  // the filter instead of re-reading node_ids (4 B) + the slot map.
  // MEASURED SLOWER at every shard size (11M: 185 vs 286 trees/s;
  // 1.375M: 999 vs 1248 — gpurun_out/sweep_slot8.log): the predicated
  // gh/bins loads break the eager unrolled pipeline, same failure mode
  // as YDFA_HIST_FILTER_SUB. Opt-in only.
  static int slot8_enabled = -1;
  if (slot8_enabled < 0) {
    const char* e = getenv("YDFA_HIST_SLOT8");
    slot8_enabled = e ? atoi(e) : 0;
  }
  // node-id-first filtered pass for single-group hinted levels: opt-in
  // (measured slower than the eager unfiltered pass; slot8 supersedes it)
  static int filter_sub_env = -1;
  if (filter_sub_env < 0) {
    const char* e = getenv("YDFA_HIST_FILTER_SUB");
    filter_sub_env = e ? atoi(e) : 0;
  }
  const int use_slot8 = (slot8_enabled && filtered_hint && !use_grp &&
                         grp_scratch != nullptr && group == n_slots &&
                         slot0 + n_slots <= 255) ? 1 : 0;
  if (use_slot8) {
    hipLaunchKernelGGL(row_group_kernel, dim3(elem_grid(N, 4096)),
                       dim3(kBlock), 0, (hipStream_t)stream, node_ids,
                       slot_map, grp_scratch, N, level_base, level_size,
                       1);
  }
  for (int s0 = 0; s0 < n_slots; s0 += group) {
    const int ng = (n_slots - s0) < group ? (n_slots - s0) : group;
    const int this_fpb = (ng == n_slots) ? fpb : 1;
    const int this_nfg = (F + this_fpb - 1) / this_fpb;
    const size_t lds = (size_t)this_fpb * ng * n_bins * 16 +
                       (lds_map ? map_bytes_full : 0);
    const int filtered =
        use_slot8 ? 2
                  : ((n_slots > group || (filtered_hint && filter_sub_env))
                         ? 1 : 0);
    const uint8_t* rg = (use_grp || use_slot8) ? grp_scratch : nullptr;
    if (swizzle) {
      const int grid_flat = ((chunks + 7) / 8) * 8 * this_nfg;
      hipLaunchKernelGGL(hist_build_lds_kernel, dim3(grid_flat),
                         dim3(threads), lds, (hipStream_t)stream, bins,
                         (const float2*)gh, node_ids, slot_map,
                         hist + (int64_t)s0 * F * n_bins * 3, N, F, n_bins,
                         level_base, level_size, slot0 + s0, ng, rg, group,
                         lds_map, filtered, this_fpb, this_nfg, chunks, 1,
                         rpb);
    } else {
      hipLaunchKernelGGL(hist_build_lds_kernel, dim3(this_nfg, chunks),
                         dim3(threads), lds, (hipStream_t)stream, bins,
                         (const float2*)gh, node_ids, slot_map,
                         hist + (int64_t)s0 * F * n_bins * 3, N, F, n_bins,
                         level_base, level_size, slot0 + s0, ng, rg, group,
                         lds_map, filtered, this_fpb, this_nfg, chunks, 0,
                         rpb);
    }
  }
}

void gpu_hist_build_gathered(const uint8_t* bins, const float* gh,
                             const int32_t* node_ids,
                             const int32_t* slot_map,
                             const int32_t* row_order, float* hist,
                             int64_t N, int F, int n_bins, int level_base,
                             int level_size, int slot0, int n_slots,
                             int64_t row_lo, int64_t row_hi, void* stream) {
  const size_t map_bytes_full = (size_t)level_size * sizeof(int32_t);
  const int lds_map = map_bytes_full <= 32 * 1024 ? 1 : 0;
  const size_t lds = (size_t)n_slots * n_bins * 16 +
                     (lds_map ? map_bytes_full : 0);
  const int64_t rows = row_hi - row_lo;
  if (rows <= 0) return;
  int chunks = (int)((2048 + F - 1) / F);
  const int64_t min_rows = 512;
  if (rows / chunks < min_rows)
    chunks = (int)((rows + min_rows - 1) / min_rows);
  if (chunks < 1) chunks = 1;
  const int64_t rpb = (rows + chunks - 1) / chunks;
  hipLaunchKernelGGL(hist_build_gathered_kernel, dim3(F, chunks),
                     dim3(kBlock), lds, (hipStream_t)stream, bins,
                     (const float2*)gh, node_ids, slot_map, row_order, hist,
                     N, F, n_bins, level_base, level_size, slot0, n_slots,
                     lds_map, row_lo, row_hi, rpb);
}

void gpu_hist_build_gathered32(const uint8_t* bins32, const float* gh,
                               const int32_t* node_ids,
                               const int32_t* slot_map,
                               const int32_t* row_order,
                               const int64_t* group_offs, float* hist,
                               const uint32_t* maskbits, int64_t N, int F,
                               int level_base, int level_size, int win0,
                               int n_groups, int64_t max_group_rows,
                               void* stream) {
  const int n_bins = kMaxBins;
  const int F32 = (F + 31) / 32;
  int chunks = 1;
  const int target_blocks = 4096;
  if ((int64_t)F32 * n_groups < target_blocks) {
    const int want = target_blocks / (F32 * (n_groups > 0 ? n_groups : 1));
    const int64_t cap = (max_group_rows + 511) / 512;
    chunks = (int)std::min<int64_t>(std::max(1, want),
                                    std::max<int64_t>(1, cap));
  }
  const size_t lds = (size_t)32 * n_bins * 16;
  int block = 1024;
  if (const char* e = std::getenv("YDFA_I32_BLOCK")) block = atoi(e);
  hipLaunchKernelGGL(hist_build_gathered32_kernel,
                     dim3(F32, chunks, n_groups), dim3(block), lds,
                     (hipStream_t)stream, bins32, (const float2*)gh,
                     node_ids, slot_map, row_order, group_offs, hist,
                     maskbits, N, F, n_bins, level_base, level_size, win0,
                     chunks);
}

void gpu_zero_hist_masked(float* hist, const uint16_t* maskbits, int F,
                          int ns, void* stream) {
  const int F16 = (F + 15) / 16;
  hipLaunchKernelGGL(zero_hist_masked_kernel, dim3(F16, 1, ns), dim3(192),
                     0, (hipStream_t)stream, hist, maskbits, F, kMaxBins,
                     ns);
}

void gpu_row_scatter(const int32_t* keys, int32_t* cursor,
                     int32_t* row_order, int64_t N, int n_keys,
                     void* stream) {
  if (n_keys > 0 && n_keys <= 8192) {
    int grid = 512;
    if (N < (int64_t)grid * kBlock)
      grid = (int)((N + kBlock - 1) / kBlock);
    if (grid < 1) grid = 1;
    hipLaunchKernelGGL(row_scatter_block_kernel, dim3(grid), dim3(kBlock),
                       (size_t)n_keys * sizeof(int), (hipStream_t)stream,
                       keys, cursor, row_order, N, n_keys);
    return;
  }
  int grid = (int)((N + kBlock - 1) / kBlock);
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(row_scatter_kernel, dim3(grid), dim3(kBlock), 0,
                     (hipStream_t)stream, keys, cursor, row_order, N);
}

void gpu_hist_build_gathered16(const uint8_t* bins16, const float* gh,
                               const int32_t* node_ids,
                               const int32_t* slot_map,
                               const int32_t* row_order,
                               const int64_t* group_offs, float* hist,
                               const uint16_t* maskbits, int64_t N, int F,
                               int level_base, int level_size, int win0,
                               int spg, int n_groups,
                               int64_t max_group_rows, void* stream) {
  const int n_bins = kMaxBins;
  const int F16 = (F + 15) / 16;
  int chunks = 1;
  const int target_blocks = 4096;
  if ((int64_t)F16 * n_groups < target_blocks) {
    const int want = target_blocks / (F16 * (n_groups > 0 ? n_groups : 1));
    const int64_t cap = (max_group_rows + 511) / 512;
    chunks = (int)std::min<int64_t>(std::max(1, want),
                                    std::max<int64_t>(1, cap));
  }
  const size_t lds = (size_t)16 * spg * n_bins * 16;
  // 1024 threads/workgroup: with spg=1 (64 KiB LDS) two workgroups fit
  // per CU -> 32 waves, hiding the scattered-load latency (measured
  // +31% over 256 threads)
  int block = 1024;
  if (const char* e = std::getenv("YDFA_I16_BLOCK")) block = atoi(e);
  if (block < 64) block = 64;
  if (block > 1024) block = 1024;
  hipLaunchKernelGGL(hist_build_gathered16_kernel,
                     dim3(F16, chunks, n_groups), dim3(block), lds,
                     (hipStream_t)stream, bins16, (const float2*)gh,
                     node_ids, slot_map, row_order, group_offs, hist,
                     maskbits, N, F, n_bins, level_base, level_size, win0,
                     spg, chunks);
}

void gpu_split_scan(const float* hist, const int32_t* abs_of_slot,
                    float* node_stats, float* best_gain_nf,
                    int32_t* best_bin_nf, int32_t* best_feat,
                    int32_t* best_bin, float* best_gain,
                    const uint8_t* feat_mask, const uint8_t* cat_flags,
                    unsigned long long* masks, const int8_t* mono,
                    float* node_bounds, int32_t* na_meanb_nf,
                    uint8_t* tree_na, int F, int n_bins, int slot0,
                    int n_slots, SplitParams sp, void* stream) {
  hipLaunchKernelGGL(split_scan_feat_kernel, dim3(n_slots, F), dim3(n_bins),
                     0, (hipStream_t)stream, hist, abs_of_slot, node_stats,
                     best_gain_nf, best_bin_nf, feat_mask, cat_flags, mono,
                     na_meanb_nf, F, n_bins, slot0, sp);
  hipLaunchKernelGGL(split_select_kernel, dim3(n_slots), dim3(kBlock), 0,
                     (hipStream_t)stream, hist, abs_of_slot, best_gain_nf,
                     best_bin_nf, node_stats, best_feat, best_bin, best_gain,
                     cat_flags, masks, mono, node_bounds, na_meanb_nf,
                     tree_na, F, n_bins, slot0, sp);
}

void gpu_plan_level(const float* node_stats, const int32_t* prev_best_feat,
                    int level_base, int level_size, int need, int use_sub,
                    int32_t* build_map, uint8_t* derived, void* stream) {
  const int grid = (level_size + kBlock - 1) / kBlock;
  hipLaunchKernelGGL(plan_level_kernel, dim3(grid), dim3(kBlock), 0,
                     (hipStream_t)stream, node_stats, prev_best_feat,
                     level_base, level_size, need, use_sub, build_map,
                     derived);
}

void gpu_subtract_hist(float* hist, const float* hist_prev,
                       const uint8_t* derived, int level_size, int F,
                       int n_bins, void* stream) {
  int gy = (int)(((int64_t)F * n_bins * 3 + kBlock - 1) / kBlock);
  if (gy > 64) gy = 64;
  hipLaunchKernelGGL(subtract_hist_kernel, dim3(level_size, gy), dim3(kBlock),
                     0, (hipStream_t)stream, hist, hist_prev, derived, F,
                     n_bins);
}

void gpu_update_node_ids(const uint8_t* bins, int32_t* node_ids,
                         const int32_t* slot_map, const int32_t* best_feat,
                         const int32_t* best_bin, const uint8_t* cat_flags,
                         const unsigned long long* masks,
                         const uint8_t* tree_na, int64_t N,
                         int level_base, int level_size, void* stream) {
  hipLaunchKernelGGL(update_node_ids_kernel, dim3(elem_grid(N, 4096)),
                     dim3(kBlock), 0, (hipStream_t)stream, bins, node_ids,
                     slot_map, best_feat, best_bin, cat_flags, masks,
                     tree_na, N,
                     level_base, level_size);
}

void gpu_leaf_values(const float* node_stats, const float* node_bounds,
                     float* leaf_values, int total_nodes, float lambda_l2,
                     float lambda_l1,
                     void* stream) {
  const int grid = (total_nodes + kBlock - 1) / kBlock;
  hipLaunchKernelGGL(leaf_values_kernel, dim3(grid), dim3(kBlock), 0,
                     (hipStream_t)stream, node_stats, node_bounds,
                     leaf_values, total_nodes, lambda_l2, lambda_l1);
}

void gpu_update_preds(float* preds, const int32_t* node_ids,
                      const float* leaf_values, int64_t N, float shrinkage,
                      void* stream) {
  hipLaunchKernelGGL(update_preds_kernel, dim3(elem_grid(N, 4096)),
                     dim3(kBlock), 0, (hipStream_t)stream, preds, node_ids,
                     leaf_values, N, shrinkage);
}

void gpu_binary_logloss(const float* preds, const float* labels, float* out2,
                        int64_t N, void* stream) {
  hipLaunchKernelGGL(binary_logloss_kernel, dim3(elem_grid(N)), dim3(kBlock),
                     0, (hipStream_t)stream, preds, labels, out2, N);
}

void gpu_pack_extract(const int32_t* feat, const int32_t* binv,
                      const float* leaf, const float* node_stats,
                      const float* gain,
                      const unsigned long long* tmasks, const uint8_t* na,
                      uint8_t* out, int T, int64_t n_mask_words,
                      void* stream) {
  const int64_t n = T > n_mask_words ? T : n_mask_words;
  hipLaunchKernelGGL(pack_extract_kernel, dim3(elem_grid(n, 256)),
                     dim3(kBlock), 0, (hipStream_t)stream, feat, binv,
                     leaf, node_stats, gain, tmasks, na, out, T,
                     n_mask_words);
}

}  // extern "C"
}  // namespace ydfa
