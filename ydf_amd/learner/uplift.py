"""Uplift tree training (capability analogue of the reference uplift
stack: learner/decision_tree/uplift.h divergence scores +
random_forest uplift support, tasks CATEGORICAL_UPLIFT /
NUMERICAL_UPLIFT).

Design: per level the existing histogram kernel runs TWICE — once over
the treatment rows ({sum w*y, sum w, count} per bin) and once over the
control rows — then split selection is a vectorized torch prefix scan
over [nodes, F, 256] computing the divergence gain
  gain = (w_l * D_l + w_r * D_r) / w - D_parent
with D one of EUCLIDEAN_DISTANCE / KULLBACK_LEIBLER / CHI_SQUARED
(uplift.h:305-352). Leaf value = E[y|treatment] - E[y|control].
Row routing reuses the same binned matrix; trees come back as HostTree
and serve through the standard flat forest.
"""
from __future__ import annotations

import math
from typing import List, Optional

import numpy as np
import torch

from ydf_amd import ops
from ydf_amd.learner.trainer import (HostTree, rf_bootstrap_weights)


def _divergence(rt: torch.Tensor, rc: torch.Tensor,
                score: str) -> torch.Tensor:
    """Per-candidate divergence (uplift.h:305): rt/rc are mean outcomes
    of treatment/control."""
    if score == "KULLBACK_LEIBLER":
        safe_rc = rc.clamp(min=1e-30)
        kl = rt * torch.log((rt / safe_rc).clamp(min=1e-30))
        out = torch.where(rc <= 0, rt / 1000.0, kl)
        return torch.where(rt <= 0, torch.zeros_like(out), out)
    if score == "CHI_SQUARED":
        chi = (rt - rc) ** 2 / rc.clamp(min=1e-30)
        return torch.where(rc <= 0, rt / 1000.0, chi)
    # EUCLIDEAN_DISTANCE (default) / CONSERVATIVE_EUCLIDEAN_DISTANCE
    return (rc - rt) ** 2


def train_uplift_forest(bins: torch.Tensor, outcome: torch.Tensor,
                        treatment: torch.Tensor, num_trees: int,
                        max_depth: int, min_examples: int,
                        min_examples_in_treatment: int,
                        split_score: str, num_candidate_features: int,
                        bootstrap: bool, seed: int,
                        weights: Optional[torch.Tensor] = None,
                        log=None) -> List[HostTree]:
    """Grows an uplift random forest. bins [F,N] u8 on any device;
    outcome f32 [N] (0/1 for categorical uplift); treatment f32 [N]
    (1 = treated)."""
    F, N = bins.shape
    dev = bins.device
    n_bins = ops.MAX_BINS
    total_nodes = (1 << (max_depth + 1)) - 1
    t_mask = (treatment > 0.5).float()
    c_mask = 1.0 - t_mask
    ghT = torch.empty((N, 2), dtype=torch.float32, device=dev)
    ghC = torch.empty((N, 2), dtype=torch.float32, device=dev)
    node_ids = torch.empty(N, dtype=torch.int32, device=dev)
    histT = torch.empty((1 << (max_depth - 1), F, n_bins, 3),
                        dtype=torch.float32, device=dev)
    histC = torch.empty_like(histT)
    arange_slots = torch.arange(1 << max_depth, dtype=torch.int32,
                                device=dev)
    trees: List[HostTree] = []
    rng = np.random.RandomState(seed)
    bins_long = None  # lazy [F,N] int64 view for torch routing

    for it in range(num_trees):
        w = torch.ones(N, dtype=torch.float32, device=dev)
        if bootstrap:
            w = rf_bootstrap_weights(seed, it, N, dev)
        if weights is not None:
            w = w * weights
        wT = w * t_mask
        wC = w * c_mask
        ghT[:, 0] = outcome * wT
        ghT[:, 1] = wT
        ghC[:, 0] = outcome * wC
        ghC[:, 1] = wC
        node_ids.zero_()
        feat_arr = np.full(total_nodes, -1, dtype=np.int32)
        bin_arr = np.zeros(total_nodes, dtype=np.int32)
        gain_arr = np.zeros(total_nodes, dtype=np.float32)
        leaf_rt = np.zeros(total_nodes, dtype=np.float64)
        leaf_rc = np.zeros(total_nodes, dtype=np.float64)
        counts_arr = np.zeros(total_nodes, dtype=np.float32)
        active = [0]
        for level in range(max_depth):
            if not active:
                break
            level_base = (1 << level) - 1
            level_size = 1 << level
            ns = len(active)
            slot_map = torch.full((level_size,), -1, dtype=torch.int32,
                                  device=dev)
            rel = torch.tensor([a - level_base for a in active],
                               dtype=torch.int64, device=dev)
            slot_map[rel] = arange_slots[:ns].clone()
            hT = histT[:ns]
            hC = histC[:ns]
            hT.zero_()
            hC.zero_()
            ops.hist_build(bins, ghT, node_ids, slot_map, hT, level_base,
                           level_size, 0, ns)
            ops.hist_build(bins, ghC, node_ids, slot_map, hC, level_base,
                           level_size, 0, ns)
            # prefix sums over bins: [ns, F, 256]
            sYT = torch.cumsum(hT[..., 0].double(), dim=2)
            sWT = torch.cumsum(hT[..., 1].double(), dim=2)
            cT = torch.cumsum(hT[..., 2].double(), dim=2)
            sYC = torch.cumsum(hC[..., 0].double(), dim=2)
            sWC = torch.cumsum(hC[..., 1].double(), dim=2)
            cC = torch.cumsum(hC[..., 2].double(), dim=2)
            totYT = sYT[..., -1:]
            totWT = sWT[..., -1:]
            totCT = cT[..., -1:]
            totYC = sYC[..., -1:]
            totWC = sWC[..., -1:]
            totCC = cC[..., -1:]
            # candidate split after bin b: left = bins <= b
            rt_l = sYT / sWT.clamp(min=1e-30)
            rc_l = sYC / sWC.clamp(min=1e-30)
            rt_r = (totYT - sYT) / (totWT - sWT).clamp(min=1e-30)
            rc_r = (totYC - sYC) / (totWC - sWC).clamp(min=1e-30)
            d_l = _divergence(rt_l, rc_l, split_score)
            d_r = _divergence(rt_r, rc_r, split_score)
            w_l = sWT + sWC
            w_r = (totWT + totWC) - w_l
            w_tot = (totWT + totWC).clamp(min=1e-30)
            rt_p = totYT / totWT.clamp(min=1e-30)
            rc_p = totYC / totWC.clamp(min=1e-30)
            d_p = _divergence(rt_p, rc_p, split_score)
            gain = (w_l * d_l + w_r * d_r) / w_tot - d_p
            n_l = cT + cC
            n_r = (totCT + totCC) - n_l
            valid = (n_l >= min_examples) & (n_r >= min_examples)
            if min_examples_in_treatment > 0:
                valid &= (cT >= min_examples_in_treatment) & \
                    ((totCT - cT) >= min_examples_in_treatment) & \
                    (cC >= min_examples_in_treatment) & \
                    ((totCC - cC) >= min_examples_in_treatment)
            valid[..., -1] = False  # no empty right side
            if 0 < num_candidate_features < F:
                # per (tree, level) feature sampling, seeded like the
                # main trainer so data-parallel ranks would agree
                rs = np.random.RandomState(
                    (seed * 1000003 + it * 8191 + level) % (1 << 31))
                keep = rs.choice(F, num_candidate_features, replace=False)
                fmask = torch.zeros(F, dtype=torch.bool, device=dev)
                fmask[torch.from_numpy(keep).to(dev)] = True
                valid &= fmask.view(1, F, 1)
            gain = torch.where(valid, gain,
                               torch.full_like(gain, -math.inf))
            flat = gain.view(ns, -1)
            best = flat.argmax(dim=1)
            best_gain = flat.gather(1, best.view(-1, 1)).view(-1)
            best_f = (best // n_bins).cpu().numpy()
            best_b = (best % n_bins).cpu().numpy()
            bg = best_gain.cpu().numpy()
            # per-node outcome stats for leaves
            rt_p_h = rt_p[..., 0, 0].cpu().numpy()
            rc_p_h = rc_p[..., 0, 0].cpu().numpy()
            cnt_h = (totCT + totCC)[..., 0, 0].cpu().numpy()
            next_active = []
            sel_feat = torch.full((level_size,), -1, dtype=torch.int32,
                                  device=dev)
            sel_bin = torch.zeros(level_size, dtype=torch.int32,
                                  device=dev)
            for s, a in enumerate(active):
                leaf_rt[a] = rt_p_h[s]
                leaf_rc[a] = rc_p_h[s]
                counts_arr[a] = cnt_h[s]
                if not np.isfinite(bg[s]) or bg[s] <= 0:
                    continue
                feat_arr[a] = best_f[s]
                bin_arr[a] = best_b[s]
                gain_arr[a] = bg[s]
                sel_feat[a - level_base] = int(best_f[s])
                sel_bin[a - level_base] = int(best_b[s])
                if level + 1 < max_depth:
                    next_active.extend((2 * a + 1, 2 * a + 2))
            ops.update_node_ids(bins, node_ids, arange_slots[:level_size],
                                sel_feat, sel_bin, level_base, level_size)
            active = next_active
        # leaf stats for the last level's children come from routing:
        # fill them with parent stats where unset (conservative), then
        # compute the tree's leaf uplift values
        leaf_value = np.zeros(total_nodes, dtype=np.float32)
        for n in range(total_nodes):
            if counts_arr[n] > 0:
                leaf_value[n] = leaf_rt[n] - leaf_rc[n]
        # children of split nodes at the deepest level never got stats;
        # compute them host-side from a final routing pass
        deep_parents = [n for n in range((1 << max_depth) - 1)
                        if feat_arr[n] >= 0
                        and counts_arr[2 * n + 1] == 0]
        if deep_parents:
            ids = node_ids.cpu().numpy()
            y_np = outcome.cpu().numpy()
            t_np = t_mask.cpu().numpy()
            w_np = w.cpu().numpy()
            for n in deep_parents:
                for child in (2 * n + 1, 2 * n + 2):
                    m = ids == child
                    wt = (w_np * t_np)[m]
                    wc = (w_np * (1 - t_np))[m]
                    yt = (y_np * w_np * t_np)[m].sum()
                    yc = (y_np * w_np * (1 - t_np))[m].sum()
                    rt = yt / max(wt.sum(), 1e-30)
                    rc = yc / max(wc.sum(), 1e-30)
                    leaf_value[child] = rt - rc
                    counts_arr[child] = m.sum()
        trees.append(HostTree(
            feat=feat_arr, bin=bin_arr, leaf_value=leaf_value,
            counts=counts_arr, max_depth=max_depth, gain=gain_arr))
        if log and (it + 1) % 100 == 0:
            log(f"trained {it + 1}/{num_trees} uplift trees")
    return trees
