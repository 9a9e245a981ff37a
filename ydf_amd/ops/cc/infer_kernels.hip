// HIP/CDNA4 batched ensemble-inference kernels (the MI355X-native serving
// engine; capability analogue of the reference's flat-node engines,
// serving/decision_forest/decision_forest_serving.cc:268-344 PredictHelper,
// redesigned example-parallel for 64-wide wavefronts with the example tile
// staged in LDS).
//
// Model layout ("flat forest", built by ydf_amd/model/forest.py):
//   feat[n]     : i32, split feature of node n, -1 => leaf
//   thr[n]      : f32, split threshold (x > thr -> right) or leaf value
//   left[n]     : i32, index of left child (right = left + 1)
//   roots[t]    : i32, root node index of tree t
//   cat_idx[n]  : i32, -1 for numerical/leaf; else index into masks of a
//                 256-bit "category goes right" bitmask (4 x u64) —
//                 capability analogue of the reference's ContainsBitmap
//                 conditions (model/decision_tree/decision_tree.proto:86-151)
// All node arrays are concatenated over trees.
#include <hip/hip_runtime.h>
#include <cstdint>
#include "common.h"

namespace ydfa {

constexpr int kTile = 256;  // examples per block

// Packed node: one 16-B gather per visit instead of three scattered loads
// (feat/thr/left in separate arrays tripled the divergent-gather count,
// which is what bounds this kernel).
struct PackedNode {
  int32_t feat;     // -1 = leaf
  float thr;        // threshold or leaf value
  int32_t left;     // left child (right = left + 1)
  int32_t cat_idx;  // -1 = numerical; else mask index
};

// Example tile staged in LDS: xs[f * kTile + tid]. The f-stride is a
// multiple of 32 banks, so per-lane-group accesses with distinct tid never
// conflict regardless of the (divergent) feature index.
__global__ void predict_forest_lds_kernel(
    const float* __restrict__ X, int64_t N, int F,
    const PackedNode* __restrict__ nodes,
    const int32_t* __restrict__ roots,
    const unsigned long long* __restrict__ masks,
    const int32_t* __restrict__ obl_ranges,
    const int32_t* __restrict__ obl_attr,
    const float* __restrict__ obl_w,
    const uint8_t* __restrict__ na_right, int has_cats,
    int tree_start, int tree_step, int n_trees, float* __restrict__ out,
    float init, float scale, int trees_per_chunk,
    float* __restrict__ partial) {
  extern __shared__ float xs[];  // [F][kTile]
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int tid = threadIdx.x;
  const int64_t n_here = min((int64_t)kTile, N - base);
  for (int idx = tid; idx < F * kTile; idx += blockDim.x) {
    const int f = idx >> 8;     // idx / kTile
    const int i = idx & 255;    // idx % kTile
    xs[idx] = (i < n_here) ? X[(int64_t)f * N + base + i] : 0.f;
  }
  __syncthreads();
  if (tid >= n_here) return;
  // tree-chunk grid for small batches: blockIdx.y covers trees
  // [t_lo, t_hi) and writes a leaf-sum partial; single-chunk launches
  // (partial == nullptr) cover the whole forest and write `out`.
  const int t_lo = blockIdx.y * trees_per_chunk;
  const int t_hi = min(t_lo + trees_per_chunk, n_trees);
  float acc = init;
  if (!has_cats && na_right == nullptr) {
    // pure-numerical fast path: 4 trees walk in parallel per thread (a
    // single walk is a chain of DEPENDENT L2 gathers), one 16-B packed
    // node load per step.
    int tt = t_lo;
    for (; tt + 4 <= t_hi; tt += 4) {
      PackedNode nd[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        nd[u] = nodes[roots[tree_start + (int64_t)(tt + u) * tree_step]];
      bool done = false;
      while (!done) {
        done = true;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if (nd[u].feat >= 0) {
            const int nx = nd[u].left +
                (xs[nd[u].feat * kTile + tid] > nd[u].thr ? 1 : 0);
            nd[u] = nodes[nx];
            done &= nd[u].feat < 0;
          }
        }
      }
#pragma unroll
      for (int u = 0; u < 4; ++u) acc += nd[u].thr;
    }
    for (; tt < t_hi; ++tt) {
      PackedNode nd = nodes[roots[tree_start + (int64_t)tt * tree_step]];
      while (nd.feat >= 0)
        nd = nodes[nd.left + (xs[nd.feat * kTile + tid] > nd.thr ? 1 : 0)];
      acc += nd.thr;
    }
  } else {
    for (int tt = t_lo; tt < t_hi; ++tt) {
      int ni = roots[tree_start + (int64_t)tt * tree_step];
      PackedNode nd = nodes[ni];
      while (nd.feat >= 0) {
        int right;
        const float xv = xs[nd.feat * kTile + tid];
        if (na_right != nullptr && (isnan(xv) || xv < 0.f) &&
            (isnan(xv) || nd.cat_idx >= 0)) {
          // missing input (NaN numerical / negative categorical code):
          // follow the stored na_value direction
          right = na_right[ni];
        } else if (nd.cat_idx >= 0) {
          int c = (int)xv;
          c = c < 0 ? 0 : (c > 255 ? 255 : c);
          right = (int)((masks[(int64_t)nd.cat_idx * 4 + (c >> 6)]
                         >> (c & 63)) & 1ull);
        } else if (nd.cat_idx <= -2) {  // oblique: sparse dot > thr
          const int oi = -(nd.cat_idx + 2);
          const int s0 = obl_ranges[2 * oi];
          const int nn = obl_ranges[2 * oi + 1];
          float dot = 0.f;
          for (int k = 0; k < nn; ++k)
            dot += obl_w[s0 + k] * xs[obl_attr[s0 + k] * kTile + tid];
          right = dot > nd.thr ? 1 : 0;
        } else {
          right = xv > nd.thr ? 1 : 0;
        }
        ni = nd.left + right;
        nd = nodes[ni];
      }
      acc += nd.thr;
    }
  }
  if (partial != nullptr)
    partial[(int64_t)blockIdx.y * N + base + tid] = acc - init;
  else
    out[base + tid] = init + (acc - init) * scale;
}

// Fallback without the LDS tile (feature count too large to stage).
__global__ void predict_forest_global_kernel(
    const float* __restrict__ X, int64_t N, int F,
    const PackedNode* __restrict__ nodes,
    const int32_t* __restrict__ roots,
    const unsigned long long* __restrict__ masks,
    const int32_t* __restrict__ obl_ranges,
    const int32_t* __restrict__ obl_attr,
    const float* __restrict__ obl_w,
    const uint8_t* __restrict__ na_right, int has_cats,
    int tree_start, int tree_step, int n_trees, float* __restrict__ out,
    float init, float scale) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride) {
    float acc = init;
    for (int tt = 0; tt < n_trees; ++tt) {
      int ni = roots[tree_start + (int64_t)tt * tree_step];
      PackedNode nd = nodes[ni];
      while (nd.feat >= 0) {
        int right;
        const float xv = X[(int64_t)nd.feat * N + k];
        if (na_right != nullptr && (isnan(xv) || xv < 0.f) &&
            (isnan(xv) || nd.cat_idx >= 0)) {
          right = na_right[ni];
        } else if (has_cats && nd.cat_idx >= 0) {
          int c = (int)xv;
          c = c < 0 ? 0 : (c > 255 ? 255 : c);
          right = (int)((masks[(int64_t)nd.cat_idx * 4 + (c >> 6)]
                         >> (c & 63)) & 1ull);
        } else if (has_cats && nd.cat_idx <= -2) {
          const int oi = -(nd.cat_idx + 2);
          const int s0 = obl_ranges[2 * oi];
          const int nn = obl_ranges[2 * oi + 1];
          float dot = 0.f;
          for (int kk = 0; kk < nn; ++kk)
            dot += obl_w[s0 + kk] * X[(int64_t)obl_attr[s0 + kk] * N + k];
          right = dot > nd.thr ? 1 : 0;
        } else {
          right = xv > nd.thr ? 1 : 0;
        }
        ni = nd.left + right;
        nd = nodes[ni];
      }
      acc += nd.thr;
    }
    out[k] = init + (acc - init) * scale;
  }
}

// ---------------------------------------------------------------------------
// QuickScorer engine (reference serving/decision_forest/
// quick_scorer_extended.h:24-61): each tree <= 64 leaves; every internal
// node carries the 64-bit mask of its LEFT subtree's leaves. For each
// condition where the example goes RIGHT (x > thr), those leaves are
// removed; the exit leaf is the lowest surviving bit. Trades the flat
// walk's 6 dependent node fetches for ~63 streamed, branch-free
// condition evaluations per tree — conditions are block-uniform reads
// (L2 broadcast), examples stay feature-major in LDS.
// ---------------------------------------------------------------------------
struct QSCond {
  int32_t feat;
  float thr;
  unsigned long long mask;
};

__global__ void qs_predict_kernel(
    const float* __restrict__ X, int64_t N, int F,
    const QSCond* __restrict__ conds,
    const int32_t* __restrict__ cond_offs,   // [T+1]
    const float* __restrict__ leaf_vals,     // [T*64]
    int n_trees, float* __restrict__ out, float init, float scale) {
  extern __shared__ float xs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int f = 0; f < F; ++f) {
    if (tid < n_here) xs[f * kTile + tid] = X[(int64_t)f * N + base + tid];
  }
  __syncthreads();
  if (tid >= n_here) return;
  float acc = 0.f;
  for (int t = 0; t < n_trees; ++t) {
    unsigned long long live = ~0ull;
    const int c1 = cond_offs[t + 1];
    for (int c = cond_offs[t]; c < c1; ++c) {
      const QSCond q = conds[c];
      if (xs[q.feat * kTile + tid] > q.thr) live &= ~q.mask;
    }
    acc += leaf_vals[(int64_t)t * 64 + (__ffsll((long long)live) - 1)];
  }
  out[base + tid] = init + acc * scale;
}

// ---------------------------------------------------------------------------
// 8-bit engine (reference serving/decision_forest/
// 8bits_numerical_features.h:82): features arrive PRE-BINNED as u8 (the
// training-side quantile bins), thresholds are bin indices, the example
// tile costs 1 byte per feature in LDS. For repeated serving of the
// same rows this skips float encode + threshold lookups entirely.
// ---------------------------------------------------------------------------
__global__ void predict_forest_binned_kernel(
    const uint8_t* __restrict__ B, int64_t N, int F,
    const PackedNode* __restrict__ nodes,
    const int32_t* __restrict__ roots, int tree_start, int tree_step,
    int n_trees, float* __restrict__ out, float init, float scale) {
  extern __shared__ uint8_t bs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int f = 0; f < F; ++f) {
    if (tid < n_here) bs[f * kTile + tid] = B[(int64_t)f * N + base + tid];
  }
  __syncthreads();
  if (tid >= n_here) return;
  float acc = init;
  for (int tt = 0; tt < n_trees; ++tt) {
    PackedNode nd = nodes[roots[tree_start + (int64_t)tt * tree_step]];
    while (nd.feat >= 0) {
      // thr field holds the split BIN as an int bit-pattern
      const int right =
          (int)bs[nd.feat * kTile + tid] > __float_as_int(nd.thr) ? 1 : 0;
      nd = nodes[nd.left + right];
    }
    acc += nd.thr;
  }
  out[base + tid] = init + (acc - init) * scale;
}

// Compact 8-byte binned node: the serving hot loop is bound by node
// fetches from L2 (depth x trees x 16 B/visit with PackedNode);
// halving the node size halves that traffic. Leaves reuse the child
// slot for the value bit-pattern.
struct Node8 {
  uint16_t feat;        // 0xFFFF = leaf
  uint16_t bin;         // split bin (binned domain)
  uint32_t left_or_val; // internal: left child; leaf: f32 bit pattern
};

__global__ void predict_forest_binned8_kernel(
    const uint8_t* __restrict__ B, int64_t N, int F,
    const Node8* __restrict__ nodes, const int32_t* __restrict__ roots,
    int tree_start, int tree_step, int n_trees, float* __restrict__ out,
    float init, float scale) {
  extern __shared__ uint8_t bs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int idx = tid; idx < F * kTile; idx += blockDim.x) {
    const int f = idx >> 8;
    const int i = idx & 255;
    bs[idx] = (i < n_here) ? B[(int64_t)f * N + base + i] : 0;
  }
  __syncthreads();
  if (tid >= n_here) return;
  float acc = init;
  // 4 trees walk in parallel per thread (same dependent-gather hiding
  // as the flat kernel) with HALF the bytes per node visit.
  int t = 0;
  for (; t + 4 <= n_trees; t += 4) {
    Node8 nd[4];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      nd[u] = nodes[roots[tree_start + (int64_t)(t + u) * tree_step]];
    bool done = false;
    while (!done) {
      done = true;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        if (nd[u].feat != 0xFFFFu) {
          const int right =
              (int)bs[nd[u].feat * kTile + tid] > (int)nd[u].bin;
          nd[u] = nodes[nd[u].left_or_val + right];
          done &= nd[u].feat == 0xFFFFu;
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) acc += __uint_as_float(nd[u].left_or_val);
  }
  for (; t < n_trees; ++t) {
    Node8 nd = nodes[roots[tree_start + (int64_t)t * tree_step]];
    while (nd.feat != 0xFFFFu) {
      const int right = (int)bs[nd.feat * kTile + tid] > (int)nd.bin;
      nd = nodes[nd.left_or_val + right];
    }
    acc += __uint_as_float(nd.left_or_val);
  }
  out[base + tid] = init + (acc - init) * scale;
}

// 4-byte node: feat (6 b, 63 = leaf) | bin (8 b) | left-child-or-
// leaf-value-index (18 b). Leaf values live in a dense side table, so
// a depth-6 visit costs 4 B + one 4-B leaf fetch per tree — ~1.7x
// less L2 traffic than the 8-B nodes. Limits: F <= 63, <= 2^18 nodes
// and leaves per forest (a 1000-tree depth-6 GBT has ~127k nodes).
__global__ void predict_forest_binned4_kernel(
    const uint8_t* __restrict__ B, int64_t N, int F,
    const uint32_t* __restrict__ nodes,
    const float* __restrict__ leaf_vals,
    const int32_t* __restrict__ roots, int tree_start, int tree_step,
    int n_trees, float* __restrict__ out, float init, float scale) {
  extern __shared__ uint8_t bs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int idx = tid; idx < F * kTile; idx += blockDim.x) {
    const int f = idx >> 8;
    const int i = idx & 255;
    bs[idx] = (i < n_here) ? B[(int64_t)f * N + base + i] : 0;
  }
  __syncthreads();
  if (tid >= n_here) return;
  float acc = init;
  int t = 0;
  for (; t + 4 <= n_trees; t += 4) {
    uint32_t nd[4];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      nd[u] = nodes[roots[tree_start + (int64_t)(t + u) * tree_step]];
    bool done = false;
    while (!done) {
      done = true;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int f = nd[u] & 63;
        if (f != 63) {
          const int right =
              (int)bs[f * kTile + tid] > (int)((nd[u] >> 6) & 255);
          nd[u] = nodes[(nd[u] >> 14) + right];
          done &= (nd[u] & 63) == 63;
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) acc += leaf_vals[nd[u] >> 14];
  }
  for (; t < n_trees; ++t) {
    uint32_t nd = nodes[roots[tree_start + (int64_t)t * tree_step]];
    while ((nd & 63) != 63) {
      const int right =
          (int)bs[(nd & 63) * kTile + tid] > (int)((nd >> 6) & 255);
      nd = nodes[(nd >> 14) + right];
    }
    acc += leaf_vals[nd >> 14];
  }
  out[base + tid] = init + (acc - init) * scale;
}

// Tree-parallel small-batch variant of the 8-byte-node engine (same
// rationale as predict_forest_binned4_tp_kernel below; used for
// wide-feature models that exceed the 4-byte packing limits).
__global__ void predict_forest_binned8_tp_kernel(
    const uint8_t* __restrict__ B, int64_t N, int F,
    const Node8* __restrict__ nodes, const int32_t* __restrict__ roots,
    int tree_start, int tree_step, int n_trees, int trees_per_chunk,
    float* __restrict__ partial) {
  extern __shared__ uint8_t bs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int idx = tid; idx < F * kTile; idx += blockDim.x) {
    const int f = idx >> 8;
    const int i = idx & 255;
    bs[idx] = (i < n_here) ? B[(int64_t)f * N + base + i] : 0;
  }
  __syncthreads();
  if (tid >= n_here) return;
  const int t_lo = blockIdx.y * trees_per_chunk;
  const int t_hi = min(t_lo + trees_per_chunk, n_trees);
  float acc = 0.f;
  int t = t_lo;
  for (; t + 4 <= t_hi; t += 4) {
    Node8 nd[4];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      nd[u] = nodes[roots[tree_start + (int64_t)(t + u) * tree_step]];
    bool done = false;
    while (!done) {
      done = true;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        if (nd[u].feat != 0xFFFFu) {
          const int right =
              (int)bs[nd[u].feat * kTile + tid] > (int)nd[u].bin;
          nd[u] = nodes[nd[u].left_or_val + right];
          done &= nd[u].feat == 0xFFFFu;
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) acc += __uint_as_float(nd[u].left_or_val);
  }
  for (; t < t_hi; ++t) {
    Node8 nd = nodes[roots[tree_start + (int64_t)t * tree_step]];
    while (nd.feat != 0xFFFFu) {
      const int right = (int)bs[nd.feat * kTile + tid] > (int)nd.bin;
      nd = nodes[nd.left_or_val + right];
    }
    acc += __uint_as_float(nd.left_or_val);
  }
  partial[(int64_t)blockIdx.y * N + base + tid] = acc;
}

// Tree-parallel variant for SMALL batches: one thread per row walks a
// serial chain of dependent node loads, so at N << 65k the row-only
// grid leaves the chip idle while each thread walks ALL trees
// (~250 x depth x L2-latency = the measured ~1 ms small-batch floor).
// Here grid.y splits the forest into tree chunks; each block writes its
// chunk's leaf-sum to a partial buffer and a tiny reduction combines
// them in FIXED chunk order (deterministic — no float atomics).
__global__ void predict_forest_binned4_tp_kernel(
    const uint8_t* __restrict__ B, int64_t N, int F,
    const uint32_t* __restrict__ nodes,
    const float* __restrict__ leaf_vals,
    const int32_t* __restrict__ roots, int tree_start, int tree_step,
    int n_trees, int trees_per_chunk, float* __restrict__ partial) {
  extern __shared__ uint8_t bs[];  // [F][kTile]
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * kTile;
  const int64_t n_here = min((int64_t)kTile, N - base);
  if (n_here <= 0) return;
  for (int idx = tid; idx < F * kTile; idx += blockDim.x) {
    const int f = idx >> 8;
    const int i = idx & 255;
    bs[idx] = (i < n_here) ? B[(int64_t)f * N + base + i] : 0;
  }
  __syncthreads();
  if (tid >= n_here) return;
  const int t_lo = blockIdx.y * trees_per_chunk;
  const int t_hi = min(t_lo + trees_per_chunk, n_trees);
  float acc = 0.f;
  int t = t_lo;
  for (; t + 4 <= t_hi; t += 4) {
    uint32_t nd[4];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      nd[u] = nodes[roots[tree_start + (int64_t)(t + u) * tree_step]];
    bool done = false;
    while (!done) {
      done = true;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int f = nd[u] & 63;
        if (f != 63) {
          const int right =
              (int)bs[f * kTile + tid] > (int)((nd[u] >> 6) & 255);
          nd[u] = nodes[(nd[u] >> 14) + right];
          done &= (nd[u] & 63) == 63;
        }
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) acc += leaf_vals[nd[u] >> 14];
  }
  for (; t < t_hi; ++t) {
    uint32_t nd = nodes[roots[tree_start + (int64_t)t * tree_step]];
    while ((nd & 63) != 63) {
      const int right =
          (int)bs[(nd & 63) * kTile + tid] > (int)((nd >> 6) & 255);
      nd = nodes[(nd >> 14) + right];
    }
    acc += leaf_vals[nd >> 14];
  }
  partial[(int64_t)blockIdx.y * N + base + tid] = acc;
}

__global__ void reduce_partials_kernel(const float* __restrict__ partial,
                                       int n_chunks, int64_t N,
                                       float* __restrict__ out,
                                       float init, float scale) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= N) return;
  float s = 0.f;
  for (int c = 0; c < n_chunks; ++c) s += partial[(int64_t)c * N + i];
  out[i] = init + s * scale;
}

__global__ void sigmoid_kernel(const float* __restrict__ in,
                               float* __restrict__ out, int64_t N) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t k = i; k < N; k += stride)
    out[k] = 1.0f / (1.0f + __expf(-in[k]));
}

// NUMERICAL_VECTOR_SEQUENCE projections — the one feature type the
// reference itself accelerated on GPU (learner/decision_tree/
// gpu.cu.cc:46-136), redesigned for wave64: one block covers a
// 256-example tile for one anchor; the anchor (+|a|^2) is staged in
// LDS; each lane scans its example's vector run computing BOTH
// condition statistics in one pass:
//   maxdot[a][i]   = max_k <vec_ik, anchor_a>       (ProjectedMoreThan)
//   negminsq[a][i] = -min_k |vec_ik - anchor_a|^2   (CloserThan)
// Empty runs stay at -3e38 (every "exists" condition false).
__global__ void vecseq_project_kernel(
    const float* __restrict__ values,   // [K][dim]
    const int64_t* __restrict__ offs,   // [N+1]
    const float* __restrict__ anchors,  // [A][dim]
    float* __restrict__ maxdot,         // [A][N]
    float* __restrict__ negminsq,       // [A][N]
    int64_t N, int dim) {
  extern __shared__ float a_s[];  // [dim]
  const int aidx = blockIdx.y;
  for (int d = threadIdx.x; d < dim; d += blockDim.x)
    a_s[d] = anchors[(int64_t)aidx * dim + d];
  __syncthreads();
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= N) return;
  float best_dot = -3.0e38f, best_neg = -3.0e38f;
  const int64_t k0 = offs[i], k1 = offs[i + 1];
  for (int64_t k = k0; k < k1; ++k) {
    const float* v = values + k * dim;
    float dot = 0.f, sq = 0.f;
    for (int d = 0; d < dim; ++d) {
      const float x = v[d];
      dot = fmaf(x, a_s[d], dot);
      const float diff = x - a_s[d];
      sq = fmaf(diff, diff, sq);
    }
    best_dot = fmaxf(best_dot, dot);
    best_neg = fmaxf(best_neg, -sq);
  }
  maxdot[(int64_t)aidx * N + i] = best_dot;
  negminsq[(int64_t)aidx * N + i] = best_neg;
}

extern "C" {

void gpu_vecseq_project(const float* values, const int64_t* offs,
                        const float* anchors, float* maxdot,
                        float* negminsq, int64_t N, int dim, int A,
                        void* stream) {
  if (A <= 0 || N <= 0) return;
  const int grid_x = (int)((N + 255) / 256);
  const size_t lds = (size_t)dim * sizeof(float);
  hipLaunchKernelGGL(vecseq_project_kernel, dim3(grid_x, A), dim3(256),
                     lds, (hipStream_t)stream, values, offs, anchors,
                     maxdot, negminsq, N, dim);
}

void gpu_predict_forest_binned(const uint8_t* B, int64_t N, int F,
                               const int32_t* packed_nodes,
                               const int32_t* roots, int tree_start,
                               int tree_step, int n_trees, float* out,
                               float init, float scale, void* stream) {
  const size_t lds = (size_t)F * kTile;
  const int grid = (int)((N + kTile - 1) / kTile);
  hipLaunchKernelGGL(predict_forest_binned_kernel, dim3(grid), dim3(kTile),
                     lds, (hipStream_t)stream, B, N, F,
                     reinterpret_cast<const PackedNode*>(packed_nodes),
                     roots, tree_start, tree_step, n_trees, out, init,
                     scale);
}

void gpu_predict_forest_binned8(const uint8_t* B, int64_t N, int F,
                                const uint32_t* nodes8,
                                const int32_t* roots, int tree_start,
                                int tree_step, int n_trees,
                                float* out, float init, float scale,
                                void* stream) {
  const size_t lds = (size_t)F * kTile;
  const int grid = (int)((N + kTile - 1) / kTile);
  hipLaunchKernelGGL(predict_forest_binned8_kernel, dim3(grid),
                     dim3(kTile), lds, (hipStream_t)stream, B, N, F,
                     reinterpret_cast<const Node8*>(nodes8), roots,
                     tree_start, tree_step, n_trees, out, init, scale);
}

void gpu_predict_forest_binned4(const uint8_t* B, int64_t N, int F,
                                const uint32_t* nodes4,
                                const float* leaf_vals,
                                const int32_t* roots, int tree_start,
                                int tree_step, int n_trees, float* out,
                                float init, float scale, void* stream) {
  const size_t lds = (size_t)F * kTile;
  const int grid = (int)((N + kTile - 1) / kTile);
  hipLaunchKernelGGL(predict_forest_binned4_kernel, dim3(grid),
                     dim3(kTile), lds, (hipStream_t)stream, B, N, F,
                     nodes4, leaf_vals, roots, tree_start, tree_step,
                     n_trees, out, init, scale);
}

void gpu_predict_forest_binned8_tp(const uint8_t* B, int64_t N, int F,
                                   const uint32_t* nodes8_u,
                                   const int32_t* roots, int tree_start,
                                   int tree_step, int n_trees,
                                   int n_chunks, float* partial,
                                   float* out, float init, float scale,
                                   void* stream) {
  const size_t lds = (size_t)F * kTile;
  const int row_tiles = (int)((N + kTile - 1) / kTile);
  const int tpc = (n_trees + n_chunks - 1) / n_chunks;
  const int chunks = (n_trees + tpc - 1) / tpc;
  hipLaunchKernelGGL(predict_forest_binned8_tp_kernel,
                     dim3(row_tiles, chunks), dim3(kTile), lds,
                     (hipStream_t)stream, B, N, F,
                     reinterpret_cast<const Node8*>(nodes8_u), roots,
                     tree_start, tree_step, n_trees, tpc, partial);
  const int rg = (int)((N + 255) / 256);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(rg), dim3(256), 0,
                     (hipStream_t)stream, partial, chunks, N, out, init,
                     scale);
}

void gpu_predict_forest_binned4_tp(const uint8_t* B, int64_t N, int F,
                                   const uint32_t* nodes4,
                                   const float* leaf_vals,
                                   const int32_t* roots, int tree_start,
                                   int tree_step, int n_trees,
                                   int n_chunks, float* partial,
                                   float* out, float init, float scale,
                                   void* stream) {
  const size_t lds = (size_t)F * kTile;
  const int row_tiles = (int)((N + kTile - 1) / kTile);
  const int tpc = (n_trees + n_chunks - 1) / n_chunks;
  const int chunks = (n_trees + tpc - 1) / tpc;
  hipLaunchKernelGGL(predict_forest_binned4_tp_kernel,
                     dim3(row_tiles, chunks), dim3(kTile), lds,
                     (hipStream_t)stream, B, N, F, nodes4, leaf_vals,
                     roots, tree_start, tree_step, n_trees, tpc,
                     partial);
  const int rg = (int)((N + 255) / 256);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(rg), dim3(256), 0,
                     (hipStream_t)stream, partial, chunks, N, out, init,
                     scale);
}

void gpu_predict_forest_qs(const float* X, int64_t N, int F,
                           const int32_t* conds, const int32_t* cond_offs,
                           const float* leaf_vals, int n_trees, float* out,
                           float init, float scale, void* stream) {
  const size_t lds = (size_t)F * kTile * sizeof(float);
  const int grid = (int)((N + kTile - 1) / kTile);
  hipLaunchKernelGGL(qs_predict_kernel, dim3(grid), dim3(kTile), lds,
                     (hipStream_t)stream, X, N, F,
                     reinterpret_cast<const QSCond*>(conds), cond_offs,
                     leaf_vals, n_trees, out, init, scale);
}

void gpu_predict_forest(const float* X, int64_t N, int F,
                        const int32_t* packed_nodes, const int32_t* roots,
                        const unsigned long long* masks,
                        const int32_t* obl_ranges, const int32_t* obl_attr,
                        const float* obl_w, const uint8_t* na_right,
                        int has_cats, int tree_start,
                        int tree_step, int n_trees, float* out, float init,
                        float scale, void* stream) {
  const PackedNode* nodes =
      reinterpret_cast<const PackedNode*>(packed_nodes);
  const size_t lds = (size_t)F * kTile * sizeof(float);
  if (lds <= 96 * 1024) {
    const int grid = (int)((N + kTile - 1) / kTile);
    hipLaunchKernelGGL(predict_forest_lds_kernel, dim3(grid), dim3(kTile), lds,
                       (hipStream_t)stream, X, N, F, nodes, roots, masks,
                       obl_ranges, obl_attr, obl_w, na_right, has_cats,
                       tree_start, tree_step, n_trees, out, init, scale,
                       n_trees, nullptr);
  } else {
    int grid = (int)((N + kTile - 1) / kTile);
    if (grid > 4096) grid = 4096;
    if (grid < 1) grid = 1;
    hipLaunchKernelGGL(predict_forest_global_kernel, dim3(grid), dim3(kTile),
                       0, (hipStream_t)stream, X, N, F, nodes, roots, masks,
                       obl_ranges, obl_attr, obl_w, na_right, has_cats,
                       tree_start, tree_step, n_trees, out, init, scale);
  }
}

void gpu_predict_forest_tp(const float* X, int64_t N, int F,
                           const int32_t* packed_nodes,
                           const int32_t* roots,
                           const unsigned long long* masks,
                           const int32_t* obl_ranges,
                           const int32_t* obl_attr, const float* obl_w,
                           const uint8_t* na_right, int has_cats,
                           int tree_start, int tree_step, int n_trees,
                           int n_chunks, float* partial, float* out,
                           float init, float scale, void* stream) {
  const PackedNode* nodes =
      reinterpret_cast<const PackedNode*>(packed_nodes);
  const size_t lds = (size_t)F * kTile * sizeof(float);
  const int row_tiles = (int)((N + kTile - 1) / kTile);
  const int tpc = (n_trees + n_chunks - 1) / n_chunks;
  const int chunks = (n_trees + tpc - 1) / tpc;
  hipLaunchKernelGGL(predict_forest_lds_kernel,
                     dim3(row_tiles, chunks), dim3(kTile), lds,
                     (hipStream_t)stream, X, N, F, nodes, roots, masks,
                     obl_ranges, obl_attr, obl_w, na_right, has_cats,
                     tree_start, tree_step, n_trees, out, 0.f, scale,
                     tpc, partial);
  const int rg = (int)((N + 255) / 256);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(rg), dim3(256), 0,
                     (hipStream_t)stream, partial, chunks, N, out, init,
                     scale);
}

void gpu_sigmoid(const float* in, float* out, int64_t N, void* stream) {
  int grid = (int)((N + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(sigmoid_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, in, out, N);
}

}  // extern "C"
}  // namespace ydfa
