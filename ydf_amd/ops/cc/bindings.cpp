// pybind11 bindings for the MI355X-native decision-forest ops.
// All tensor arguments are passed as raw device/host addresses
// (torch.Tensor.data_ptr()) plus shapes; `stream` is
// torch.cuda.current_stream().cuda_stream. This keeps the extension free of
// torch headers (fast hipcc builds, no hipify involvement anywhere).
#include <pybind11/pybind11.h>

#include <cstdint>

#include "common.h"

namespace py = pybind11;
using ydfa::SplitParams;

extern "C" {
// train_kernels.hip
void gpu_bin_data(const float*, const float*, uint8_t*, int64_t, int, int,
                  int, void*);
void gpu_grad_hess(const float*, const float*, float*, int64_t, int, void*);
void gpu_grad_hess_softmax(const float*, const float*, float*, int64_t, int,
                           int, void*);
void gpu_hist_build(const uint8_t*, const float*, const int32_t*,
                    const int32_t*, float*, int64_t, int, int, int, int, int,
                    int, int, uint8_t*, void*);
void gpu_weighted_target(const float*, const float*, float*, int64_t, void*);
void gpu_hist_build_gathered32(const uint8_t*, const float*,
                               const int32_t*, const int32_t*,
                               const int32_t*, const int64_t*, float*,
                               const uint32_t*, int64_t, int, int, int,
                               int, int, int64_t, void*);
void gpu_zero_hist_masked(float*, const uint16_t*, int, int, void*);
void gpu_row_scatter(const int32_t*, int32_t*, int32_t*, int64_t, int,
                     void*);
void gpu_hist_build_gathered16(const uint8_t*, const float*,
                               const int32_t*, const int32_t*,
                               const int32_t*, const int64_t*, float*,
                               const uint16_t*, int64_t, int, int, int,
                               int, int, int, int64_t, void*);
void gpu_hist_build_gathered(const uint8_t*, const float*, const int32_t*,
                             const int32_t*, const int32_t*, float*, int64_t,
                             int, int, int, int, int, int, int64_t, int64_t,
                             void*);
void gpu_split_scan(const float*, const int32_t*, float*, float*, int32_t*,
                    int32_t*, int32_t*, float*, const uint8_t*,
                    const uint8_t*, unsigned long long*, const int8_t*,
                    float*, int32_t*, uint8_t*, int, int, int, int,
                    SplitParams, void*);
void gpu_plan_level(const float*, const int32_t*, int, int, int, int,
                    int32_t*, uint8_t*, void*);
void gpu_subtract_hist(float*, const float*, const uint8_t*, int, int, int,
                       void*);
void gpu_update_node_ids(const uint8_t*, int32_t*, const int32_t*,
                         const int32_t*, const int32_t*, const uint8_t*,
                         const unsigned long long*, const uint8_t*,
                         int64_t, int, int, void*);
void gpu_leaf_values(const float*, const float*, float*, int, float,
                     float, void*);
void gpu_update_preds(float*, const int32_t*, const float*, int64_t, float,
                      void*);
void gpu_binary_logloss(const float*, const float*, float*, int64_t, void*);
void gpu_pack_extract(const int32_t*, const int32_t*, const float*,
                      const float*, const float*, const unsigned long long*,
                      const uint8_t*, uint8_t*, int, int64_t, void*);
// infer_kernels.hip
void gpu_predict_forest(const float*, int64_t, int, const int32_t*,
                        const int32_t*, const unsigned long long*,
                        const int32_t*, const int32_t*, const float*,
                        const uint8_t*, int,
                        int, int, int, float*, float, float, void*);
void gpu_sigmoid(const float*, float*, int64_t, void*);
void gpu_predict_forest_binned(const uint8_t*, int64_t, int,
                               const int32_t*, const int32_t*, int, int,
                               int, float*, float, float, void*);
void gpu_predict_forest_qs(const float*, int64_t, int, const int32_t*,
                           const int32_t*, const float*, int, float*,
                           float, float, void*);
void gpu_vecseq_project(const float*, const int64_t*, const float*,
                        float*, float*, int64_t, int, int, void*);
void gpu_predict_forest_binned8(const uint8_t*, int64_t, int,
                                const uint32_t*, const int32_t*, int,
                                int, int, float*, float, float, void*);
void gpu_predict_forest_binned4(const uint8_t*, int64_t, int,
                                const uint32_t*, const float*,
                                const int32_t*, int, int, int, float*,
                                float, float, void*);
void gpu_predict_forest_binned4_tp(const uint8_t*, int64_t, int,
                                   const uint32_t*, const float*,
                                   const int32_t*, int, int, int, int,
                                   float*, float*, float, float, void*);
void gpu_predict_forest_tp(const float*, int64_t, int, const int32_t*,
                           const int32_t*, const unsigned long long*,
                           const int32_t*, const int32_t*, const float*,
                           const uint8_t*, int, int, int, int, int,
                           float*, float*, float, float, void*);
void gpu_predict_forest_binned8_tp(const uint8_t*, int64_t, int,
                                   const uint32_t*, const int32_t*, int,
                                   int, int, int, float*, float*, float,
                                   float, void*);
// cpu_ops.cpp
void cpu_bin_data(const float*, const float*, uint8_t*, int64_t, int, int,
                  int);
void cpu_grad_hess(const float*, const float*, float*, int64_t, int);
void cpu_grad_hess_softmax(const float*, const float*, float*, int64_t, int,
                           int);
void cpu_hist_build(const uint8_t*, const float*, const int32_t*,
                    const int32_t*, float*, int64_t, int, int, int, int, int,
                    int);
void cpu_weighted_target(const float*, const float*, float*, int64_t);
void cpu_split_scan(const float*, const int32_t*, float*, float*, int32_t*,
                    int32_t*, int32_t*, float*, const uint8_t*,
                    const uint8_t*, unsigned long long*, const int8_t*,
                    float*, int32_t*, uint8_t*, int, int, int, int,
                    SplitParams);
void cpu_plan_level(const float*, const int32_t*, int, int, int, int,
                    int32_t*, uint8_t*);
void cpu_subtract_hist(float*, const float*, const uint8_t*, int, int, int);
void cpu_update_node_ids(const uint8_t*, int32_t*, const int32_t*,
                         const int32_t*, const int32_t*, const uint8_t*,
                         const unsigned long long*, const uint8_t*,
                         int64_t, int, int);
void cpu_leaf_values(const float*, const float*, float*, int, float,
                     float);
void cpu_update_preds(float*, const int32_t*, const float*, int64_t, float);
void cpu_binary_logloss(const float*, const float*, float*, int64_t);
void cpu_predict_forest(const float*, int64_t, int, const int32_t*,
                        const float*, const int32_t*, const int32_t*,
                        const int32_t*, const unsigned long long*,
                        const int32_t*, const int32_t*, const float*,
                        const uint8_t*, int,
                        int, int, float*, float, float);
void cpu_tree_shap(const float*, int64_t, int, const int32_t*, const float*,
                   const int32_t*, const int32_t*, const unsigned long long*,
                   const int32_t*, const int32_t*, const float*,
                   const uint8_t*,
                   const float*, const int32_t*, int, int, int, float, float,
                   float*);
void cpu_forest_expected_value(const int32_t*, const float*, const int32_t*,
                               const float*, const int32_t*, int, int, int,
                               float, double*);
}

namespace {
template <typename T>
T* P(uintptr_t p) {
  return reinterpret_cast<T*>(p);
}
SplitParams MakeSP(float lambda_l2, float min_hessian, int min_examples,
                   float min_gain, float cat_smooth, float lambda_l1,
                   int na_mode) {
  SplitParams sp;
  sp.lambda_l2 = lambda_l2;
  sp.lambda_l1 = lambda_l1;
  sp.min_hessian = min_hessian;
  sp.min_examples = min_examples;
  sp.min_gain = min_gain;
  sp.cat_smooth = cat_smooth;
  sp.na_mode = na_mode;
  return sp;
}
}  // namespace

PYBIND11_MODULE(_ydf_ops, m) {
  m.doc() = "MI355X-native decision forest ops (HIP/gfx950 + CPU)";
  m.attr("max_bins") = ydfa::kMaxBins;
  const auto nogil = py::call_guard<py::gil_scoped_release>();

  // --- GPU ---
  m.def("gpu_bin_data",
        [](uintptr_t x, uintptr_t bnd, uintptr_t out, int64_t N, int F,
           int n_cuts, int na_to_255, uintptr_t stream) {
          gpu_bin_data(P<float>(x), P<float>(bnd), P<uint8_t>(out), N, F,
                       n_cuts, na_to_255, (void*)stream);
        },
        nogil);
  m.def("gpu_grad_hess",
        [](uintptr_t preds, uintptr_t labels, uintptr_t gh, int64_t N,
           int loss, uintptr_t stream) {
          gpu_grad_hess(P<float>(preds), P<float>(labels), P<float>(gh), N,
                        loss, (void*)stream);
        },
        nogil);
  m.def("gpu_grad_hess_softmax",
        [](uintptr_t preds, uintptr_t labels, uintptr_t gh, int64_t N,
           int n_classes, int cls, uintptr_t stream) {
          gpu_grad_hess_softmax(P<float>(preds), P<float>(labels),
                                P<float>(gh), N, n_classes, cls,
                                (void*)stream);
        },
        nogil);
  m.def("gpu_hist_build",
        [](uintptr_t bins, uintptr_t gh, uintptr_t node_ids,
           uintptr_t slot_map, uintptr_t hist, int64_t N, int F, int n_bins,
           int level_base, int level_size, int slot0, int n_slots,
           int filtered_hint, uintptr_t grp_scratch, uintptr_t stream) {
          gpu_hist_build(P<uint8_t>(bins), P<float>(gh), P<int32_t>(node_ids),
                         P<int32_t>(slot_map), P<float>(hist), N, F, n_bins,
                         level_base, level_size, slot0, n_slots,
                         filtered_hint, P<uint8_t>(grp_scratch),
                         (void*)stream);
        },
        nogil);
  m.def("gpu_hist_build_gathered",
        [](uintptr_t bins, uintptr_t gh, uintptr_t node_ids,
           uintptr_t slot_map, uintptr_t row_order, uintptr_t hist,
           int64_t N, int F, int n_bins, int level_base, int level_size,
           int slot0, int n_slots, int64_t row_lo, int64_t row_hi,
           uintptr_t stream) {
          gpu_hist_build_gathered(
              P<uint8_t>(bins), P<float>(gh), P<int32_t>(node_ids),
              P<int32_t>(slot_map), P<int32_t>(row_order), P<float>(hist),
              N, F, n_bins, level_base, level_size, slot0, n_slots, row_lo,
              row_hi, (void*)stream);
        },
        nogil);
  m.def("gpu_hist_build_gathered32",
        [](uintptr_t bins32, uintptr_t gh, uintptr_t node_ids,
           uintptr_t slot_map, uintptr_t row_order, uintptr_t group_offs,
           uintptr_t hist, uintptr_t maskbits, int64_t N, int F,
           int level_base, int level_size, int win0, int n_groups,
           int64_t max_group_rows, uintptr_t stream) {
          gpu_hist_build_gathered32(
              P<uint8_t>(bins32), P<float>(gh), P<int32_t>(node_ids),
              P<int32_t>(slot_map), P<int32_t>(row_order),
              P<int64_t>(group_offs), P<float>(hist),
              P<uint32_t>(maskbits), N, F, level_base, level_size, win0,
              n_groups, max_group_rows, (void*)stream);
        },
        nogil);
  m.def("gpu_zero_hist_masked",
        [](uintptr_t hist, uintptr_t maskbits, int F, int ns,
           uintptr_t stream) {
          gpu_zero_hist_masked(P<float>(hist), P<uint16_t>(maskbits), F,
                               ns, (void*)stream);
        },
        nogil);
  m.def("gpu_row_scatter",
        [](uintptr_t keys, uintptr_t cursor, uintptr_t row_order,
           int64_t N, int n_keys, uintptr_t stream) {
          gpu_row_scatter(P<int32_t>(keys), P<int32_t>(cursor),
                          P<int32_t>(row_order), N, n_keys,
                          (void*)stream);
        },
        nogil);
  m.def("gpu_hist_build_gathered16",
        [](uintptr_t bins16, uintptr_t gh, uintptr_t node_ids,
           uintptr_t slot_map, uintptr_t row_order, uintptr_t group_offs,
           uintptr_t hist, uintptr_t maskbits, int64_t N, int F,
           int level_base, int level_size, int win0, int spg, int n_groups,
           int64_t max_group_rows, uintptr_t stream) {
          gpu_hist_build_gathered16(
              P<uint8_t>(bins16), P<float>(gh), P<int32_t>(node_ids),
              P<int32_t>(slot_map), P<int32_t>(row_order),
              P<int64_t>(group_offs), P<float>(hist),
              P<uint16_t>(maskbits), N, F, level_base,
              level_size, win0, spg, n_groups, max_group_rows,
              (void*)stream);
        },
        nogil);
  m.def("gpu_weighted_target",
        [](uintptr_t labels, uintptr_t weights, uintptr_t gh, int64_t N,
           uintptr_t stream) {
          gpu_weighted_target(P<float>(labels), P<float>(weights), P<float>(gh),
                              N, (void*)stream);
        },
        nogil);
  m.def("gpu_split_scan",
        [](uintptr_t hist, uintptr_t abs_of_slot, uintptr_t node_stats,
           uintptr_t best_gain_nf, uintptr_t best_bin_nf, uintptr_t best_feat,
           uintptr_t best_bin, uintptr_t best_gain, uintptr_t feat_mask,
           uintptr_t cat_flags, uintptr_t masks, uintptr_t mono,
           uintptr_t node_bounds, int F, int n_bins, int slot0,
           int n_slots, float lambda_l2, float min_hessian, int min_examples,
           float min_gain, float cat_smooth, float lambda_l1,
           uintptr_t na_meanb_nf, uintptr_t tree_na, int na_mode,
           uintptr_t stream) {
          gpu_split_scan(P<float>(hist), P<int32_t>(abs_of_slot),
                         P<float>(node_stats), P<float>(best_gain_nf),
                         P<int32_t>(best_bin_nf), P<int32_t>(best_feat),
                         P<int32_t>(best_bin), P<float>(best_gain),
                         P<uint8_t>(feat_mask), P<uint8_t>(cat_flags),
                         P<unsigned long long>(masks), P<int8_t>(mono),
                         P<float>(node_bounds), P<int32_t>(na_meanb_nf),
                         P<uint8_t>(tree_na), F, n_bins, slot0, n_slots,
                         MakeSP(lambda_l2, min_hessian, min_examples,
                                min_gain, cat_smooth, lambda_l1, na_mode),
                         (void*)stream);
        },
        nogil);
  m.def("gpu_plan_level",
        [](uintptr_t node_stats, uintptr_t prev_best_feat, int level_base,
           int level_size, int need, int use_sub, uintptr_t build_map,
           uintptr_t derived, uintptr_t stream) {
          gpu_plan_level(P<float>(node_stats), P<int32_t>(prev_best_feat),
                         level_base, level_size, need, use_sub,
                         P<int32_t>(build_map), P<uint8_t>(derived),
                         (void*)stream);
        },
        nogil);
  m.def("gpu_subtract_hist",
        [](uintptr_t hist, uintptr_t hist_prev, uintptr_t derived,
           int level_size, int F, int n_bins, uintptr_t stream) {
          gpu_subtract_hist(P<float>(hist), P<float>(hist_prev),
                            P<uint8_t>(derived), level_size, F, n_bins,
                            (void*)stream);
        },
        nogil);
  m.def("gpu_update_node_ids",
        [](uintptr_t bins, uintptr_t node_ids, uintptr_t slot_map,
           uintptr_t best_feat, uintptr_t best_bin, uintptr_t cat_flags,
           uintptr_t masks, uintptr_t tree_na, int64_t N, int level_base,
           int level_size, uintptr_t stream) {
          gpu_update_node_ids(P<uint8_t>(bins), P<int32_t>(node_ids),
                              P<int32_t>(slot_map), P<int32_t>(best_feat),
                              P<int32_t>(best_bin), P<uint8_t>(cat_flags),
                              P<unsigned long long>(masks),
                              P<uint8_t>(tree_na), N, level_base,
                              level_size, (void*)stream);
        },
        nogil);
  m.def("gpu_leaf_values",
        [](uintptr_t node_stats, uintptr_t node_bounds, uintptr_t leaf_values,
           int total_nodes, float lambda_l2, float lambda_l1,
           uintptr_t stream) {
          gpu_leaf_values(P<float>(node_stats), P<float>(node_bounds),
                          P<float>(leaf_values), total_nodes, lambda_l2,
                          lambda_l1,
                          (void*)stream);
        },
        nogil);
  m.def("gpu_pack_extract",
        [](uintptr_t feat, uintptr_t binv, uintptr_t leaf,
           uintptr_t node_stats, uintptr_t gain, uintptr_t tmasks,
           uintptr_t na, uintptr_t out, int T, int64_t n_mask_words,
           uintptr_t stream) {
          gpu_pack_extract(P<int32_t>(feat), P<int32_t>(binv),
                           P<float>(leaf), P<float>(node_stats),
                           P<float>(gain),
                           P<unsigned long long>(tmasks), P<uint8_t>(na),
                           P<uint8_t>(out), T, n_mask_words,
                           (void*)stream);
        },
        nogil);
  m.def("gpu_update_preds",
        [](uintptr_t preds, uintptr_t node_ids, uintptr_t leaf_values,
           int64_t N, float shrinkage, uintptr_t stream) {
          gpu_update_preds(P<float>(preds), P<int32_t>(node_ids),
                           P<float>(leaf_values), N, shrinkage,
                           (void*)stream);
        },
        nogil);
  m.def("gpu_binary_logloss",
        [](uintptr_t preds, uintptr_t labels, uintptr_t out2, int64_t N,
           uintptr_t stream) {
          gpu_binary_logloss(P<float>(preds), P<float>(labels), P<float>(out2),
                             N, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest",
        [](uintptr_t X, int64_t N, int F, uintptr_t packed_nodes,
           uintptr_t roots, uintptr_t masks, uintptr_t obl_ranges,
           uintptr_t obl_attr, uintptr_t obl_w, uintptr_t na_right,
           int has_cats, int tree_start,
           int tree_step, int n_trees, uintptr_t out, float init,
           float scale, uintptr_t stream) {
          gpu_predict_forest(P<float>(X), N, F, P<int32_t>(packed_nodes),
                             P<int32_t>(roots),
                             P<unsigned long long>(masks),
                             P<int32_t>(obl_ranges), P<int32_t>(obl_attr),
                             P<float>(obl_w), P<uint8_t>(na_right),
                             has_cats,
                             tree_start, tree_step, n_trees, P<float>(out),
                             init, scale, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_binned",
        [](uintptr_t B, int64_t N, int F, uintptr_t packed_nodes,
           uintptr_t roots, int tree_start, int tree_step, int n_trees,
           uintptr_t out, float init, float scale, uintptr_t stream) {
          gpu_predict_forest_binned(P<uint8_t>(B), N, F,
                                    P<int32_t>(packed_nodes),
                                    P<int32_t>(roots), tree_start,
                                    tree_step, n_trees, P<float>(out),
                                    init, scale, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_qs",
        [](uintptr_t X, int64_t N, int F, uintptr_t conds,
           uintptr_t cond_offs, uintptr_t leaf_vals, int n_trees,
           uintptr_t out, float init, float scale, uintptr_t stream) {
          gpu_predict_forest_qs(P<float>(X), N, F, P<int32_t>(conds),
                                P<int32_t>(cond_offs), P<float>(leaf_vals),
                                n_trees, P<float>(out), init, scale,
                                (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_binned4",
        [](uintptr_t B, int64_t N, int F, uintptr_t nodes4,
           uintptr_t leaf_vals, uintptr_t roots, int tree_start,
           int tree_step, int n_trees, uintptr_t out, float init,
           float scale, uintptr_t stream) {
          gpu_predict_forest_binned4(P<uint8_t>(B), N, F,
                                     P<uint32_t>(nodes4),
                                     P<float>(leaf_vals),
                                     P<int32_t>(roots), tree_start,
                                     tree_step, n_trees, P<float>(out),
                                     init, scale, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_tp",
        [](uintptr_t X, int64_t N, int F, uintptr_t nodes,
           uintptr_t roots, uintptr_t masks, uintptr_t obl_ranges,
           uintptr_t obl_attr, uintptr_t obl_w, uintptr_t na_right,
           int has_cats, int tree_start, int tree_step, int n_trees,
           int n_chunks, uintptr_t partial, uintptr_t out, float init,
           float scale, uintptr_t stream) {
          gpu_predict_forest_tp(
              P<float>(X), N, F, P<int32_t>(nodes), P<int32_t>(roots),
              P<unsigned long long>(masks), P<int32_t>(obl_ranges),
              P<int32_t>(obl_attr), P<float>(obl_w),
              P<uint8_t>(na_right), has_cats, tree_start, tree_step,
              n_trees, n_chunks, P<float>(partial), P<float>(out),
              init, scale, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_binned8_tp",
        [](uintptr_t B, int64_t N, int F, uintptr_t nodes8,
           uintptr_t roots, int tree_start, int tree_step, int n_trees,
           int n_chunks, uintptr_t partial, uintptr_t out, float init,
           float scale, uintptr_t stream) {
          gpu_predict_forest_binned8_tp(
              P<uint8_t>(B), N, F, P<uint32_t>(nodes8),
              P<int32_t>(roots), tree_start, tree_step, n_trees,
              n_chunks, P<float>(partial), P<float>(out), init, scale,
              (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_binned4_tp",
        [](uintptr_t B, int64_t N, int F, uintptr_t nodes4,
           uintptr_t leaf_vals, uintptr_t roots, int tree_start,
           int tree_step, int n_trees, int n_chunks, uintptr_t partial,
           uintptr_t out, float init, float scale, uintptr_t stream) {
          gpu_predict_forest_binned4_tp(
              P<uint8_t>(B), N, F, P<uint32_t>(nodes4),
              P<float>(leaf_vals), P<int32_t>(roots), tree_start,
              tree_step, n_trees, n_chunks, P<float>(partial),
              P<float>(out), init, scale, (void*)stream);
        },
        nogil);
  m.def("gpu_predict_forest_binned8",
        [](uintptr_t B, int64_t N, int F, uintptr_t nodes8,
           uintptr_t roots, int tree_start, int tree_step, int n_trees,
           uintptr_t out, float init, float scale, uintptr_t stream) {
          gpu_predict_forest_binned8(P<uint8_t>(B), N, F,
                                     P<uint32_t>(nodes8),
                                     P<int32_t>(roots), tree_start,
                                     tree_step, n_trees,
                                     P<float>(out), init, scale,
                                     (void*)stream);
        },
        nogil);
  m.def("gpu_vecseq_project",
        [](uintptr_t values, uintptr_t offs, uintptr_t anchors,
           uintptr_t maxdot, uintptr_t negminsq, int64_t N, int dim,
           int A, uintptr_t stream) {
          gpu_vecseq_project(P<float>(values), P<int64_t>(offs),
                             P<float>(anchors), P<float>(maxdot),
                             P<float>(negminsq), N, dim, A,
                             (void*)stream);
        },
        nogil);
  m.def("gpu_sigmoid",
        [](uintptr_t in, uintptr_t out, int64_t N, uintptr_t stream) {
          gpu_sigmoid(P<float>(in), P<float>(out), N, (void*)stream);
        },
        nogil);

  // --- CPU ---
  m.def("cpu_bin_data",
        [](uintptr_t x, uintptr_t bnd, uintptr_t out, int64_t N, int F,
           int n_cuts, int na_to_255) {
          cpu_bin_data(P<float>(x), P<float>(bnd), P<uint8_t>(out), N, F,
                       n_cuts, na_to_255);
        },
        nogil);
  m.def("cpu_grad_hess",
        [](uintptr_t preds, uintptr_t labels, uintptr_t gh, int64_t N,
           int loss) {
          cpu_grad_hess(P<float>(preds), P<float>(labels), P<float>(gh), N,
                        loss);
        },
        nogil);
  m.def("cpu_grad_hess_softmax",
        [](uintptr_t preds, uintptr_t labels, uintptr_t gh, int64_t N,
           int n_classes, int cls) {
          cpu_grad_hess_softmax(P<float>(preds), P<float>(labels),
                                P<float>(gh), N, n_classes, cls);
        },
        nogil);
  m.def("cpu_hist_build",
        [](uintptr_t bins, uintptr_t gh, uintptr_t node_ids,
           uintptr_t slot_map, uintptr_t hist, int64_t N, int F, int n_bins,
           int level_base, int level_size, int slot0, int n_slots) {
          cpu_hist_build(P<uint8_t>(bins), P<float>(gh), P<int32_t>(node_ids),
                         P<int32_t>(slot_map), P<float>(hist), N, F, n_bins,
                         level_base, level_size, slot0, n_slots);
        },
        nogil);
  m.def("cpu_weighted_target",
        [](uintptr_t labels, uintptr_t weights, uintptr_t gh, int64_t N) {
          cpu_weighted_target(P<float>(labels), P<float>(weights),
                              P<float>(gh), N);
        },
        nogil);
  m.def("cpu_split_scan",
        [](uintptr_t hist, uintptr_t abs_of_slot, uintptr_t node_stats,
           uintptr_t best_gain_nf, uintptr_t best_bin_nf, uintptr_t best_feat,
           uintptr_t best_bin, uintptr_t best_gain, uintptr_t feat_mask,
           uintptr_t cat_flags, uintptr_t masks, uintptr_t mono,
           uintptr_t node_bounds, int F, int n_bins, int slot0,
           int n_slots, float lambda_l2, float min_hessian, int min_examples,
           float min_gain, float cat_smooth, float lambda_l1,
           uintptr_t na_meanb_nf, uintptr_t tree_na, int na_mode) {
          cpu_split_scan(P<float>(hist), P<int32_t>(abs_of_slot),
                         P<float>(node_stats), P<float>(best_gain_nf),
                         P<int32_t>(best_bin_nf), P<int32_t>(best_feat),
                         P<int32_t>(best_bin), P<float>(best_gain),
                         P<uint8_t>(feat_mask), P<uint8_t>(cat_flags),
                         P<unsigned long long>(masks), P<int8_t>(mono),
                         P<float>(node_bounds), P<int32_t>(na_meanb_nf),
                         P<uint8_t>(tree_na), F, n_bins, slot0, n_slots,
                         MakeSP(lambda_l2, min_hessian, min_examples,
                                min_gain, cat_smooth, lambda_l1, na_mode));
        },
        nogil);
  m.def("cpu_plan_level",
        [](uintptr_t node_stats, uintptr_t prev_best_feat, int level_base,
           int level_size, int need, int use_sub, uintptr_t build_map,
           uintptr_t derived) {
          cpu_plan_level(P<float>(node_stats), P<int32_t>(prev_best_feat),
                         level_base, level_size, need, use_sub,
                         P<int32_t>(build_map), P<uint8_t>(derived));
        },
        nogil);
  m.def("cpu_subtract_hist",
        [](uintptr_t hist, uintptr_t hist_prev, uintptr_t derived,
           int level_size, int F, int n_bins) {
          cpu_subtract_hist(P<float>(hist), P<float>(hist_prev),
                            P<uint8_t>(derived), level_size, F, n_bins);
        },
        nogil);
  m.def("cpu_update_node_ids",
        [](uintptr_t bins, uintptr_t node_ids, uintptr_t slot_map,
           uintptr_t best_feat, uintptr_t best_bin, uintptr_t cat_flags,
           uintptr_t masks, uintptr_t tree_na, int64_t N, int level_base,
           int level_size) {
          cpu_update_node_ids(P<uint8_t>(bins), P<int32_t>(node_ids),
                              P<int32_t>(slot_map), P<int32_t>(best_feat),
                              P<int32_t>(best_bin), P<uint8_t>(cat_flags),
                              P<unsigned long long>(masks),
                              P<uint8_t>(tree_na), N, level_base,
                              level_size);
        },
        nogil);
  m.def("cpu_leaf_values",
        [](uintptr_t node_stats, uintptr_t node_bounds, uintptr_t leaf_values,
           int total_nodes, float lambda_l2, float lambda_l1) {
          cpu_leaf_values(P<float>(node_stats), P<float>(node_bounds),
                          P<float>(leaf_values), total_nodes, lambda_l2,
                          lambda_l1);
        },
        nogil);
  m.def("cpu_update_preds",
        [](uintptr_t preds, uintptr_t node_ids, uintptr_t leaf_values,
           int64_t N, float shrinkage) {
          cpu_update_preds(P<float>(preds), P<int32_t>(node_ids),
                           P<float>(leaf_values), N, shrinkage);
        },
        nogil);
  m.def("cpu_binary_logloss",
        [](uintptr_t preds, uintptr_t labels, uintptr_t out2, int64_t N) {
          cpu_binary_logloss(P<float>(preds), P<float>(labels), P<float>(out2),
                             N);
        },
        nogil);
  m.def("cpu_tree_shap",
        [](uintptr_t X, int64_t N, int F, uintptr_t feat, uintptr_t thr,
           uintptr_t left, uintptr_t cat_idx, uintptr_t masks,
           uintptr_t obl_ranges, uintptr_t obl_attr, uintptr_t obl_w,
           uintptr_t na_right,
           uintptr_t cover, uintptr_t roots, int tree_start, int tree_step,
           int n_trees, float scale, float init, uintptr_t phi_out) {
          cpu_tree_shap(P<float>(X), N, F, P<int32_t>(feat), P<float>(thr),
                        P<int32_t>(left), P<int32_t>(cat_idx),
                        P<unsigned long long>(masks), P<int32_t>(obl_ranges),
                        P<int32_t>(obl_attr), P<float>(obl_w),
                        P<uint8_t>(na_right),
                        P<float>(cover),
                        P<int32_t>(roots), tree_start, tree_step, n_trees,
                        scale, init, P<float>(phi_out));
        },
        nogil);
  m.def("cpu_forest_expected_value",
        [](uintptr_t feat, uintptr_t thr, uintptr_t left, uintptr_t cover,
           uintptr_t roots, int tree_start, int tree_step, int n_trees,
           float scale) {
          double out = 0.0;
          cpu_forest_expected_value(P<int32_t>(feat), P<float>(thr),
                                    P<int32_t>(left), P<float>(cover),
                                    P<int32_t>(roots), tree_start, tree_step,
                                    n_trees, scale, &out);
          return out;
        });
  m.def("cpu_predict_forest",
        [](uintptr_t X, int64_t N, int F, uintptr_t feat, uintptr_t thr,
           uintptr_t left, uintptr_t roots, uintptr_t cat_idx,
           uintptr_t masks, uintptr_t obl_ranges, uintptr_t obl_attr,
           uintptr_t obl_w, uintptr_t na_right, int tree_start,
           int tree_step, int n_trees,
           uintptr_t out, float init, float scale) {
          cpu_predict_forest(P<float>(X), N, F, P<int32_t>(feat),
                             P<float>(thr), P<int32_t>(left),
                             P<int32_t>(roots), P<int32_t>(cat_idx),
                             P<unsigned long long>(masks),
                             P<int32_t>(obl_ranges), P<int32_t>(obl_attr),
                             P<float>(obl_w), P<uint8_t>(na_right),
                             tree_start,
                             tree_step, n_trees, P<float>(out), init, scale);
        },
        nogil);
}
