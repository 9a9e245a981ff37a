"""Out-of-core (>HBM) GBT training over an on-disk binned cache.

Reference analogue: ShardedSamplingTrain
(learner/gradient_boosted_trees/gradient_boosted_trees.cc:655) and the
distributed dataset-cache training loop (dataset_cache.h:15-58).

MI355X design: one fused pass over the row chunks per tree LEVEL —
(data-parallel: each rank owns the chunk subset c % world == rank and
the per-level histograms all-reduce over RCCL, so one on-disk cache
feeds all 8 GPUs of a node) —
each chunk is uploaded (H2D on GPU, memory-mapped on CPU), the
previous level's routing is applied to that chunk's node ids,
gradients are recomputed from the chunk's resident predictions, and
the chunk's contribution is accumulated into the level histograms.
Device memory holds ONE chunk + the per-level histogram tensor +
per-chunk row state; the dataset itself can exceed HBM. Split
selection runs once per level on the accumulated histograms — the
same split_scan kernel as the in-memory path, so semantics match up
to floating-point accumulation order across chunks.
"""
from __future__ import annotations

from typing import List

import numpy as np
import torch

from ydf_amd import ops
from ydf_amd.learner.trainer import HostTree


def train_gbt_streaming(cache, cfg, device: torch.device,
                        log=None) -> (List[HostTree], float):
    """Trains cfg.num_trees boosting iterations over a DatasetCache.

    Supports: binary classification (loss 1) / regression (loss 2),
    numerical + categorical (<=256 vocab) features, depth-wise growth.
    Returns (trees, init_prediction)."""
    import torch.distributed as _td

    distributed = _td.is_available() and _td.is_initialized()
    rank = _td.get_rank() if distributed else 0
    world = _td.get_world_size() if distributed else 1

    def allreduce(t):
        if not distributed:
            return
        if t.is_cuda and _td.get_backend() == "gloo":
            h = t.cpu()
            _td.all_reduce(h, op=_td.ReduceOp.SUM)
            t.copy_(h)
        else:
            _td.all_reduce(t, op=_td.ReduceOp.SUM)

    my_chunks = [c for c in range(cache.n_chunks) if c % world == rank]
    F = cache.n_features
    N = cache.n_rows
    n_bins = 256
    D = cfg.max_depth
    total_nodes = (1 << (D + 1)) - 1

    cat_flags_t = None
    if cache.cat_flags.any():
        cat_flags_t = torch.from_numpy(
            cache.cat_flags.astype(np.uint8)).to(device)

    # --- init prediction from a streaming pass over OWNED labels ----
    s = torch.zeros(1, dtype=torch.float64, device=device)
    for c in my_chunks:
        _, labels = cache.chunk(c)
        s += float(np.asarray(labels, dtype=np.float64).sum())
    allreduce(s)
    s = float(s.item())
    if cfg.loss == 1:
        p = min(max(s / N, 1e-6), 1 - 1e-6)
        init = float(np.log(p / (1 - p)))
    else:
        init = float(s / N)

    # --- per-OWNED-chunk persistent host state ----------------------
    chunk_meta = {}
    preds_h = {}
    node_ids_h = {}
    for c in my_chunks:
        rows = min(cache.chunk_rows, N - c * cache.chunk_rows)
        chunk_meta[c] = rows
        preds_h[c] = np.full(rows, init, dtype=np.float32)
        node_ids_h[c] = np.zeros(rows, dtype=np.int32)

    # --- device/host buffers ----------------------------------------
    widest = 1 << (D - 1)
    hist = torch.zeros((widest, F, n_bins, 3), dtype=torch.float32,
                       device=device)
    node_stats = torch.zeros((total_nodes, 3), dtype=torch.float32,
                             device=device)
    bg_nf = torch.empty((widest, F), dtype=torch.float32, device=device)
    bb_nf = torch.empty((widest, F), dtype=torch.int32, device=device)
    best_feat = torch.empty(widest, dtype=torch.int32, device=device)
    best_bin = torch.empty(widest, dtype=torch.int32, device=device)
    best_gain = torch.empty(widest, dtype=torch.float32, device=device)
    leaf_vals = torch.empty(total_nodes, dtype=torch.float32,
                            device=device)
    tree_masks = torch.zeros((total_nodes, 4), dtype=torch.int64,
                             device=device) if cat_flags_t is not None \
        else None
    arange = torch.arange(widest, dtype=torch.int32, device=device)

    trees: List[HostTree] = []
    tree_feat_all = np.full(total_nodes, -1, dtype=np.int32)
    tree_bin_all = np.zeros(total_nodes, dtype=np.int32)

    from ydf_amd.ops import split_scan

    for t in range(cfg.num_trees):
        node_stats.zero_()
        if tree_masks is not None:
            tree_masks.zero_()
        tree_feat_all.fill(-1)
        tree_bin_all.fill(0)
        for c in my_chunks:
            node_ids_h[c].fill(0)
        prev_bf = prev_bb = None
        for level in range(D):
            level_base = (1 << level) - 1
            level_size = 1 << level
            hist_view = hist[:level_size]
            hist_view.zero_()
            slot_map = arange[:level_size]
            for c in my_chunks:
                bins_np, labels_np = cache.chunk(c)
                rows = chunk_meta[c]
                bins_c = torch.from_numpy(
                    np.ascontiguousarray(bins_np)).to(device)
                nid_c = torch.from_numpy(node_ids_h[c]).to(device)
                if level > 0:
                    # apply the PREVIOUS level's routing to this chunk
                    ops.update_node_ids(
                        bins_c, nid_c, slot_map[: level_size // 2],
                        prev_bf, prev_bb,
                        (1 << (level - 1)) - 1, level_size // 2,
                        cat_flags=cat_flags_t, masks=tree_masks)
                    node_ids_h[c][:] = nid_c.cpu().numpy()
                preds_c = torch.from_numpy(preds_h[c]).to(device)
                labels_c = torch.from_numpy(
                    np.ascontiguousarray(labels_np)).to(device)
                gh_c = torch.empty((rows, 2), dtype=torch.float32,
                                   device=device)
                ops.grad_hess(preds_c, labels_c, gh_c, cfg.loss)
                ops.hist_build(bins_c, gh_c, nid_c, slot_map,
                               hist_view, level_base, level_size, 0,
                               level_size)
                del bins_c, nid_c, preds_c, labels_c, gh_c
            allreduce(hist_view)
            split_scan(hist_view, slot_map + level_base, node_stats,
                       bg_nf[:level_size], bb_nf[:level_size],
                       best_feat, best_bin, best_gain, 0, level_size,
                       cfg.lambda_l2, cfg.min_hessian, cfg.min_examples,
                       cfg.min_gain, cat_flags=cat_flags_t,
                       masks=tree_masks, cat_smooth=cfg.cat_smooth,
                       lambda_l1=cfg.lambda_l1)
            bf = best_feat[:level_size].cpu().numpy()
            bb = best_bin[:level_size].cpu().numpy()
            tree_feat_all[level_base:level_base + level_size] = bf
            tree_bin_all[level_base:level_base + level_size] = bb
            prev_bf = best_feat[:level_size].clone()
            prev_bb = best_bin[:level_size].clone()
        # leaf values + prediction update (final routing pass)
        ops.leaf_values(node_stats, leaf_vals, cfg.lambda_l2,
                        lambda_l1=cfg.lambda_l1)
        last_base = (1 << (D - 1)) - 1
        last_size = 1 << (D - 1)
        for c in my_chunks:
            bins_np, _ = cache.chunk(c)
            bins_c = torch.from_numpy(
                np.ascontiguousarray(bins_np)).to(device)
            nid_c = torch.from_numpy(node_ids_h[c]).to(device)
            ops.update_node_ids(bins_c, nid_c, arange[:last_size],
                                prev_bf, prev_bb, last_base, last_size,
                                cat_flags=cat_flags_t, masks=tree_masks)
            node_ids_h[c][:] = nid_c.cpu().numpy()
            preds_c = torch.from_numpy(preds_h[c]).to(device)
            ops.update_preds(preds_c, nid_c, leaf_vals, cfg.shrinkage)
            preds_h[c][:] = preds_c.cpu().numpy()
            del bins_c, nid_c, preds_c
        trees.append(HostTree(
            feat=tree_feat_all.copy(),
            bin=tree_bin_all.copy(),
            leaf_value=leaf_vals.cpu().numpy().copy(),
            counts=node_stats[:, 2].cpu().numpy().copy(),
            max_depth=D,
            masks=tree_masks.cpu().numpy().view(np.uint64).copy()
            if tree_masks is not None else None,
            gain=np.zeros(total_nodes, dtype=np.float32)))
        if log and (t + 1) % 10 == 0:
            log(f"streaming GBT: {t + 1}/{cfg.num_trees} trees")
    return trees, init
