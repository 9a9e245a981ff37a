"""Torch-tensor front end for the MI355X-native decision-forest ops.

Dispatches each op to the hand-written HIP/gfx950 kernel (device tensors) or
the C++ CPU implementation (host tensors). The HIP path is mandatory on GPU:
if the extension is missing while a CUDA device is visible, import fails
loudly instead of silently falling back.
"""
from __future__ import annotations

import torch

try:
    from ydf_amd import _ydf_ops as _C
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "ydf_amd._ydf_ops native extension not built. Run "
        "`python tools/build_ext.py` (hipcc, gfx950). Refusing to run "
        "without the native kernels."
    ) from e

MAX_BINS = _C.max_bins


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _chk(t: torch.Tensor, dtype, name: str) -> None:
    assert t.dtype == dtype, f"{name}: expected {dtype}, got {t.dtype}"
    assert t.is_contiguous(), f"{name}: must be contiguous"


def bin_data(x: torch.Tensor, boundaries: torch.Tensor,
             out: torch.Tensor, na_to_255: bool = False) -> torch.Tensor:
    """x [F,N] f32, boundaries [F,n_cuts] f32 (ascending) -> out [F,N]
    u8. na_to_255: NaN lands in the reserved bin 255
    (LOCAL_IMPUTATION)."""
    F, N = x.shape
    n_cuts = boundaries.shape[1]
    _chk(x, torch.float32, "x")
    _chk(boundaries, torch.float32, "boundaries")
    _chk(out, torch.uint8, "out")
    if x.is_cuda:
        _C.gpu_bin_data(x.data_ptr(), boundaries.data_ptr(), out.data_ptr(),
                        N, F, n_cuts, 1 if na_to_255 else 0, _stream())
    else:
        _C.cpu_bin_data(x.data_ptr(), boundaries.data_ptr(), out.data_ptr(),
                        N, F, n_cuts, 1 if na_to_255 else 0)
    return out


def grad_hess(preds: torch.Tensor, labels: torch.Tensor, gh: torch.Tensor,
              loss: int) -> torch.Tensor:
    """preds/labels [N] f32 -> gh [N,2] f32 interleaved {g,h}."""
    N = preds.numel()
    if preds.is_cuda:
        _C.gpu_grad_hess(preds.data_ptr(), labels.data_ptr(), gh.data_ptr(),
                         N, loss, _stream())
    else:
        _C.cpu_grad_hess(preds.data_ptr(), labels.data_ptr(), gh.data_ptr(),
                         N, loss)
    return gh


def grad_hess_softmax(preds: torch.Tensor, labels: torch.Tensor,
                      gh: torch.Tensor, n_classes: int,
                      cls: int) -> torch.Tensor:
    """preds [C,N] f32 (logits), labels [N] f32 (class idx) -> gh for `cls`."""
    N = labels.numel()
    if preds.is_cuda:
        _C.gpu_grad_hess_softmax(preds.data_ptr(), labels.data_ptr(),
                                 gh.data_ptr(), N, n_classes, cls, _stream())
    else:
        _C.cpu_grad_hess_softmax(preds.data_ptr(), labels.data_ptr(),
                                 gh.data_ptr(), N, n_classes, cls)
    return gh


def weighted_target(labels: torch.Tensor, weights, gh: torch.Tensor):
    """RF/CART target: g=-w*y, h=w. weights may be None (unit)."""
    N = labels.numel()
    wp = weights.data_ptr() if weights is not None else 0
    if labels.is_cuda:
        _C.gpu_weighted_target(labels.data_ptr(), wp, gh.data_ptr(), N,
                               _stream())
    else:
        _C.cpu_weighted_target(labels.data_ptr(), wp, gh.data_ptr(), N)
    return gh


def pack_extract(feat: torch.Tensor, binv: torch.Tensor,
                 leaf: torch.Tensor, node_stats: torch.Tensor,
                 gain: torch.Tensor, tmasks, na, out: torch.Tensor,
                 T: int, n_mask_words: int):
    """Packs the per-tree extraction arrays into `out` (u8 staging
    buffer) in ONE kernel launch; layout documented at
    pack_extract_kernel (train_kernels.hip). GPU only."""
    _C.gpu_pack_extract(
        feat.data_ptr(), binv.data_ptr(), leaf.data_ptr(),
        node_stats.data_ptr(), gain.data_ptr(),
        tmasks.data_ptr() if tmasks is not None else 0,
        na.data_ptr() if na is not None else 0,
        out.data_ptr(), T, n_mask_words, _stream())


def hist_build(bins: torch.Tensor, gh: torch.Tensor, node_ids: torch.Tensor,
               slot_map: torch.Tensor, hist: torch.Tensor, level_base: int,
               level_size: int, slot0: int, n_slots: int,
               filtered_hint: bool = False, grp_scratch=None):
    """bins [F,N] u8; hist [n_slots,F,n_bins,3] f32 (pre-zeroed, base=slot0).

    slot_map [level_size] i32 maps level-relative node -> slot (-1 closed).
    filtered_hint: many rows will not match (e.g. histogram-subtraction
    levels build only the smaller children) -> use the node-id-first path."""
    F, N = bins.shape
    n_bins = hist.shape[2]
    if bins.is_cuda:
        _C.gpu_hist_build(bins.data_ptr(), gh.data_ptr(), node_ids.data_ptr(),
                          slot_map.data_ptr(), hist.data_ptr(), N, F, n_bins,
                          level_base, level_size, slot0, n_slots,
                          1 if filtered_hint else 0,
                          grp_scratch.data_ptr() if grp_scratch is not None
                          else 0, _stream())
    else:
        _C.cpu_hist_build(bins.data_ptr(), gh.data_ptr(), node_ids.data_ptr(),
                          slot_map.data_ptr(), hist.data_ptr(), N, F, n_bins,
                          level_base, level_size, slot0, n_slots)
    return hist


def hist_build_gathered16(bins16: torch.Tensor, gh: torch.Tensor,
                          node_ids: torch.Tensor, slot_map: torch.Tensor,
                          row_order: torch.Tensor,
                          group_offs: torch.Tensor, hist: torch.Tensor,
                          N: int, F: int, level_base: int,
                          level_size: int, win0: int, spg: int,
                          n_groups: int, max_group_rows: int,
                          maskbits=None):
    """Feature-interleaved partitioned histograms (GPU only): bins16
    [ceil(F/16), N, 16] u8; one uint4 load covers 16 features. Groups of
    `spg` slots map to blockIdx.z with row ranges from group_offs.
    maskbits [n_slots, ceil(F/16)] u16: per-slot feature-sampling bits
    (skips loads/atomics for unsampled features)."""
    assert bins16.is_cuda
    mb = maskbits.data_ptr() if maskbits is not None else 0
    _C.gpu_hist_build_gathered16(
        bins16.data_ptr(), gh.data_ptr(), node_ids.data_ptr(),
        slot_map.data_ptr(), row_order.data_ptr(), group_offs.data_ptr(),
        hist.data_ptr(), mb, N, F, level_base, level_size, win0, spg,
        n_groups, max_group_rows, _stream())
    return hist


def hist_build_gathered32(bins32: torch.Tensor, gh: torch.Tensor,
                          node_ids: torch.Tensor, slot_map: torch.Tensor,
                          row_order: torch.Tensor,
                          group_offs: torch.Tensor, hist: torch.Tensor,
                          N: int, F: int, level_base: int,
                          level_size: int, win0: int, n_groups: int,
                          max_group_rows: int, maskbits=None):
    """32-feature interleaved partitioned histograms (one slot/block);
    maskbits [ns, ceil(F/32)] u32."""
    mb = maskbits.data_ptr() if maskbits is not None else 0
    _C.gpu_hist_build_gathered32(
        bins32.data_ptr(), gh.data_ptr(), node_ids.data_ptr(),
        slot_map.data_ptr(), row_order.data_ptr(), group_offs.data_ptr(),
        hist.data_ptr(), mb, N, F, level_base, level_size, win0,
        n_groups, max_group_rows, _stream())
    return hist


def pack_bins32(bins: torch.Tensor) -> torch.Tensor:
    """[F, N] u8 -> [ceil(F/32), N, 32] u8 interleaved copy."""
    F, N = bins.shape
    F32 = (F + 31) // 32
    padded = torch.zeros((F32 * 32, N), dtype=torch.uint8,
                         device=bins.device)
    padded[:F] = bins
    return padded.view(F32, 32, N).permute(0, 2, 1).contiguous()


def zero_hist_masked(hist: torch.Tensor, maskbits: torch.Tensor,
                     F: int, ns: int):
    """Zeroes only the (slot, sampled-feature) histogram cells."""
    _C.gpu_zero_hist_masked(hist.data_ptr(), maskbits.data_ptr(), F, ns,
                            _stream())
    return hist


def row_scatter(keys: torch.Tensor, cursor: torch.Tensor,
                row_order: torch.Tensor):
    """Counting-sort scatter: row_order[cursor[keys[r]]++] = r.
    cursor length = number of distinct keys (block-local LDS counting
    when it fits)."""
    _C.gpu_row_scatter(keys.data_ptr(), cursor.data_ptr(),
                       row_order.data_ptr(), keys.numel(),
                       int(cursor.numel()), _stream())
    return row_order


def pack_bins16(bins: torch.Tensor) -> torch.Tensor:
    """[F, N] u8 -> [ceil(F/16), N, 16] u8 interleaved copy."""
    F, N = bins.shape
    F16 = (F + 15) // 16
    padded = torch.zeros((F16 * 16, N), dtype=torch.uint8,
                         device=bins.device)
    padded[:F] = bins
    return padded.view(F16, 16, N).permute(0, 2, 1).contiguous()


def split_scan(hist: torch.Tensor, abs_of_slot: torch.Tensor,
               node_stats: torch.Tensor, best_gain_nf: torch.Tensor,
               best_bin_nf: torch.Tensor, best_feat: torch.Tensor,
               best_bin: torch.Tensor, best_gain: torch.Tensor, slot0: int,
               n_slots: int, lambda_l2: float, min_hessian: float,
               min_examples: int, min_gain: float, feat_mask=None,
               cat_flags=None, masks=None, cat_smooth: float = 1.0,
               mono=None, node_bounds=None, lambda_l1: float = 0.0,
               na_meanb_nf=None, tree_na=None):
    F = hist.shape[1]
    n_bins = hist.shape[2]
    mp = feat_mask.data_ptr() if feat_mask is not None else 0
    cf = cat_flags.data_ptr() if cat_flags is not None else 0
    mk = masks.data_ptr() if masks is not None else 0
    mn = mono.data_ptr() if mono is not None else 0
    nb = node_bounds.data_ptr() if node_bounds is not None else 0
    args = (hist.data_ptr(), abs_of_slot.data_ptr(), node_stats.data_ptr(),
            best_gain_nf.data_ptr(), best_bin_nf.data_ptr(),
            best_feat.data_ptr(), best_bin.data_ptr(), best_gain.data_ptr(),
            mp, cf, mk, mn, nb, F, n_bins, slot0, n_slots, lambda_l2,
            min_hessian, min_examples, min_gain, cat_smooth, lambda_l1,
            na_meanb_nf.data_ptr() if na_meanb_nf is not None else 0,
            tree_na.data_ptr() if tree_na is not None else 0,
            1 if na_meanb_nf is not None else 0)
    if hist.is_cuda:
        _C.gpu_split_scan(*args, _stream())
    else:
        _C.cpu_split_scan(*args)


def hist_build_gathered(bins, gh, node_ids, slot_map, row_order, hist,
                        level_base, level_size, slot0, n_slots, row_lo,
                        row_hi):
    """Deep-level histogram build over a contiguous row-partitioned range
    of `row_order` (GPU only)."""
    F, N = bins.shape
    n_bins = hist.shape[2]
    _C.gpu_hist_build_gathered(
        bins.data_ptr(), gh.data_ptr(), node_ids.data_ptr(),
        slot_map.data_ptr(), row_order.data_ptr(), hist.data_ptr(), N, F,
        n_bins, level_base, level_size, slot0, n_slots, row_lo, row_hi,
        _stream())
    return hist


def plan_level(node_stats: torch.Tensor, prev_best_feat: torch.Tensor,
               level_base: int, level_size: int, need: int, use_sub: bool,
               build_map: torch.Tensor, derived: torch.Tensor):
    """Dense-mode device planning: build/derive/skip per level node."""
    args = (node_stats.data_ptr(), prev_best_feat.data_ptr(), level_base,
            level_size, need, 1 if use_sub else 0, build_map.data_ptr(),
            derived.data_ptr())
    if node_stats.is_cuda:
        _C.gpu_plan_level(*args, _stream())
    else:
        _C.cpu_plan_level(*args)


def subtract_hist(hist: torch.Tensor, hist_prev: torch.Tensor,
                  derived: torch.Tensor, level_size: int):
    F, n_bins = hist.shape[1], hist.shape[2]
    args = (hist.data_ptr(), hist_prev.data_ptr(), derived.data_ptr(),
            level_size, F, n_bins)
    if hist.is_cuda:
        _C.gpu_subtract_hist(*args, _stream())
    else:
        _C.cpu_subtract_hist(*args)


def update_node_ids(bins: torch.Tensor, node_ids: torch.Tensor,
                    slot_map: torch.Tensor, best_feat: torch.Tensor,
                    best_bin: torch.Tensor, level_base: int, level_size: int,
                    cat_flags=None, masks=None, tree_na=None):
    F, N = bins.shape
    cf = cat_flags.data_ptr() if cat_flags is not None else 0
    mk = masks.data_ptr() if masks is not None else 0
    na = tree_na.data_ptr() if tree_na is not None else 0
    if bins.is_cuda:
        _C.gpu_update_node_ids(bins.data_ptr(), node_ids.data_ptr(),
                               slot_map.data_ptr(), best_feat.data_ptr(),
                               best_bin.data_ptr(), cf, mk, na, N,
                               level_base, level_size, _stream())
    else:
        _C.cpu_update_node_ids(bins.data_ptr(), node_ids.data_ptr(),
                               slot_map.data_ptr(), best_feat.data_ptr(),
                               best_bin.data_ptr(), cf, mk, na, N,
                               level_base, level_size)


def leaf_values(node_stats: torch.Tensor, out: torch.Tensor,
                lambda_l2: float, node_bounds=None,
                lambda_l1: float = 0.0):
    total = out.numel()
    nb = node_bounds.data_ptr() if node_bounds is not None else 0
    if node_stats.is_cuda:
        _C.gpu_leaf_values(node_stats.data_ptr(), nb, out.data_ptr(), total,
                           lambda_l2, lambda_l1, _stream())
    else:
        _C.cpu_leaf_values(node_stats.data_ptr(), nb, out.data_ptr(), total,
                           lambda_l2, lambda_l1)
    return out


def update_preds(preds: torch.Tensor, node_ids: torch.Tensor,
                 leaf_vals: torch.Tensor, shrinkage: float):
    N = preds.numel()
    if preds.is_cuda:
        _C.gpu_update_preds(preds.data_ptr(), node_ids.data_ptr(),
                            leaf_vals.data_ptr(), N, shrinkage, _stream())
    else:
        _C.cpu_update_preds(preds.data_ptr(), node_ids.data_ptr(),
                            leaf_vals.data_ptr(), N, shrinkage)


def binary_logloss(preds: torch.Tensor, labels: torch.Tensor,
                   out2: torch.Tensor):
    """Accumulates {loss_sum, n_correct} into out2 [2] f32 (pre-zeroed)."""
    N = preds.numel()
    if preds.is_cuda:
        _C.gpu_binary_logloss(preds.data_ptr(), labels.data_ptr(),
                              out2.data_ptr(), N, _stream())
    else:
        _C.cpu_binary_logloss(preds.data_ptr(), labels.data_ptr(),
                              out2.data_ptr(), N)
    return out2


def pack_forest_nodes(feat: torch.Tensor, thr: torch.Tensor,
                      left: torch.Tensor, cat_idx: torch.Tensor
                      ) -> torch.Tensor:
    """Interleaves node arrays into 16-B {feat, thr, left, cat_idx}
    structs (one gather per visit on GPU)."""
    packed = torch.empty((feat.numel(), 4), dtype=torch.int32,
                         device=feat.device)
    packed[:, 0] = feat
    packed[:, 1] = thr.view(torch.int32)
    packed[:, 2] = left
    packed[:, 3] = cat_idx
    return packed


def predict_forest(X: torch.Tensor, feat: torch.Tensor, thr: torch.Tensor,
                   left: torch.Tensor, roots: torch.Tensor, out: torch.Tensor,
                   tree_start: int = 0, tree_step: int = 1,
                   n_trees: int = -1, init: float = 0.0, scale: float = 1.0,
                   cat_idx=None, masks=None, packed=None,
                   obl_ranges=None, obl_attr=None, obl_w=None,
                   na_right=None):
    """Flat-forest batch inference. X [F,N] f32, out [N] f32. On GPU,
    pass `packed` (pack_forest_nodes output) for the fast path. Oblique
    nodes (cat_idx <= -2) read obl_ranges [n,2] i32 / obl_attr i32 /
    obl_w f32."""
    F, N = X.shape
    if n_trees < 0:
        n_trees = roots.numel()
    mk = masks.data_ptr() if masks is not None else 0
    orr = obl_ranges.data_ptr() if obl_ranges is not None else 0
    oa = obl_attr.data_ptr() if obl_attr is not None else 0
    ow = obl_w.data_ptr() if obl_w is not None else 0
    nr = na_right.data_ptr() if na_right is not None else 0
    special = cat_idx is not None or obl_ranges is not None \
        or na_right is not None
    if X.is_cuda:
        if packed is None:
            ci = cat_idx if cat_idx is not None else torch.full(
                (feat.numel(),), -1, dtype=torch.int32, device=X.device)
            packed = pack_forest_nodes(feat, thr, left, ci)
        row_tiles = (N + 255) // 256
        lds_ok = F * 256 * 4 <= 96 * 1024
        if lds_ok and row_tiles * 4 < 512 and n_trees >= 64:
            # small batch: tree-parallel grid (deterministic per-chunk
            # partials; see predict_forest_binned4)
            n_chunks = max(1, min(512 // max(row_tiles, 1),
                                  n_trees // 16))
            partial = torch.empty((n_chunks, N), dtype=torch.float32,
                                  device=X.device)
            _C.gpu_predict_forest_tp(
                X.data_ptr(), N, F, packed.data_ptr(),
                roots.data_ptr(), mk, orr, oa, ow, nr,
                1 if special else 0, tree_start, tree_step, n_trees,
                n_chunks, partial.data_ptr(), out.data_ptr(), init,
                scale, _stream())
            return out
        _C.gpu_predict_forest(X.data_ptr(), N, F, packed.data_ptr(),
                              roots.data_ptr(), mk, orr, oa, ow, nr,
                              1 if special else 0, tree_start,
                              tree_step, n_trees, out.data_ptr(), init,
                              scale, _stream())
    else:
        ci = cat_idx.data_ptr() if cat_idx is not None else 0
        _C.cpu_predict_forest(X.data_ptr(), N, F, feat.data_ptr(),
                              thr.data_ptr(), left.data_ptr(),
                              roots.data_ptr(), ci, mk, orr, oa, ow, nr,
                              tree_start, tree_step,
                              n_trees, out.data_ptr(), init, scale)
    return out


def predict_forest_binned(B: torch.Tensor, packed: torch.Tensor,
                          roots: torch.Tensor, out: torch.Tensor,
                          tree_start: int = 0, tree_step: int = 1,
                          n_trees: int = -1, init: float = 0.0,
                          scale: float = 1.0):
    """8-bit engine (GPU): B [F,N] u8 pre-binned features; `packed`
    holds the split BIN index in the thr slot
    (see model.forest binned packing)."""
    assert B.is_cuda
    F, N = B.shape
    if n_trees < 0:
        n_trees = roots.numel()
    _C.gpu_predict_forest_binned(B.data_ptr(), N, F, packed.data_ptr(),
                                 roots.data_ptr(), tree_start, tree_step,
                                 n_trees, out.data_ptr(), init, scale,
                                 _stream())
    return out


def predict_forest_qs(X: torch.Tensor, conds: torch.Tensor,
                      cond_offs: torch.Tensor, leaf_vals: torch.Tensor,
                      out: torch.Tensor, init: float = 0.0,
                      scale: float = 1.0):
    """QuickScorer batch inference (GPU; numerical conditions,
    <= 64 leaves/tree). conds [C,4] i32, cond_offs [T+1] i32,
    leaf_vals [T,64] f32 (build_quickscorer output)."""
    assert X.is_cuda
    F, N = X.shape
    T = cond_offs.numel() - 1
    _C.gpu_predict_forest_qs(X.data_ptr(), N, F, conds.data_ptr(),
                             cond_offs.data_ptr(), leaf_vals.data_ptr(),
                             T, out.data_ptr(), init, scale, _stream())
    return out


def sigmoid(x: torch.Tensor, out: torch.Tensor):
    if x.is_cuda:
        _C.gpu_sigmoid(x.data_ptr(), out.data_ptr(), x.numel(), _stream())
    else:
        torch.sigmoid(x, out=out)
    return out


def vecseq_project(values: torch.Tensor, offs: torch.Tensor,
                   anchors: torch.Tensor):
    """NUMERICAL_VECTOR_SEQUENCE projections (reference gpu.cu.cc:46-136
    redesigned for wave64 + LDS anchors): returns (maxdot [A,N],
    negminsq [A,N]) f32. CPU falls back to the vectorized numpy twin."""
    N = offs.numel() - 1
    A = anchors.shape[0]
    dim = anchors.shape[1] if A else 1
    if values.is_cuda:
        maxdot = torch.empty((A, N), dtype=torch.float32,
                             device=values.device)
        negminsq = torch.empty((A, N), dtype=torch.float32,
                               device=values.device)
        maxdot.fill_(-3.0e38)
        negminsq.fill_(-3.0e38)
        _C.gpu_vecseq_project(values.data_ptr(), offs.data_ptr(),
                              anchors.data_ptr(), maxdot.data_ptr(),
                              negminsq.data_ptr(), N, dim, A, _stream())
        return maxdot, negminsq
    from ydf_amd.dataset.vecseq import project_numpy

    md, ns = project_numpy(values.numpy(), offs.numpy(), anchors.numpy())
    return torch.from_numpy(md), torch.from_numpy(ns)


def predict_forest_binned8(B: torch.Tensor, packed8: torch.Tensor,
                           roots: torch.Tensor, out: torch.Tensor,
                           init: float = 0.0, tree_start: int = 0,
                           tree_step: int = 1, n_trees: int = -1):
    """Compact-node 8-bit engine (GPU): B [F,N] u8 pre-binned features;
    packed8 [n_nodes, 2] u32 (pack_binned8_nodes output, leaf values
    pre-scaled). Half the node bytes per visit of the 16-B engines.
    tree_start/tree_step stride class trees for multi-output models."""
    assert B.is_cuda
    F, N = B.shape
    if n_trees < 0:
        n_trees = roots.numel()
    row_tiles = (N + 255) // 256
    if row_tiles * 4 < 512 and n_trees >= 64:
        # small batch: tree-parallel grid (see predict_forest_binned4)
        n_chunks = max(1, min(512 // max(row_tiles, 1), n_trees // 16))
        partial = torch.empty((n_chunks, N), dtype=torch.float32,
                              device=B.device)
        _C.gpu_predict_forest_binned8_tp(
            B.data_ptr(), N, F, packed8.data_ptr(), roots.data_ptr(),
            tree_start, tree_step, n_trees, n_chunks,
            partial.data_ptr(), out.data_ptr(), init, 1.0, _stream())
        return out
    _C.gpu_predict_forest_binned8(B.data_ptr(), N, F,
                                  packed8.data_ptr(), roots.data_ptr(),
                                  tree_start, tree_step, n_trees,
                                  out.data_ptr(), init, 1.0, _stream())
    return out


def predict_forest_binned4(B: torch.Tensor, nodes4: torch.Tensor,
                           leaf_vals: torch.Tensor, roots: torch.Tensor,
                           out: torch.Tensor, init: float = 0.0,
                           tree_start: int = 0, tree_step: int = 1,
                           n_trees: int = -1):
    """4-byte-node binned engine (GPU): see pack_binned4_nodes."""
    assert B.is_cuda
    F, N = B.shape
    if n_trees < 0:
        n_trees = roots.numel()
    row_tiles = (N + 255) // 256
    if row_tiles * 4 < 512 and n_trees >= 64:
        # small batch: a row-only grid underfills the chip and each
        # thread serially walks the whole forest (dependent L2 loads).
        # Split the forest across grid.y into deterministic partials.
        n_chunks = max(1, min(512 // max(row_tiles, 1), n_trees // 16))
        partial = torch.empty((n_chunks, N), dtype=torch.float32,
                              device=B.device)
        _C.gpu_predict_forest_binned4_tp(
            B.data_ptr(), N, F, nodes4.data_ptr(),
            leaf_vals.data_ptr(), roots.data_ptr(), tree_start,
            tree_step, n_trees, n_chunks, partial.data_ptr(),
            out.data_ptr(), init, 1.0, _stream())
        return out
    _C.gpu_predict_forest_binned4(B.data_ptr(), N, F, nodes4.data_ptr(),
                                  leaf_vals.data_ptr(), roots.data_ptr(),
                                  tree_start, tree_step, n_trees,
                                  out.data_ptr(), init, 1.0, _stream())
    return out
