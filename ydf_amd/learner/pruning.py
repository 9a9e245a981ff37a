"""CART validation-set pruning (reference learner/cart/cart.h:44 +
cart.cc pruning: a node becomes a leaf when the leaf's validation loss is
not worse than its subtree's)."""
from __future__ import annotations

import numpy as np

from ydf_amd.dataset.dataspec import Task


def prune_tree(tree, Xv: np.ndarray, yv: np.ndarray, bnd: np.ndarray,
               cat_feats, task: Task) -> int:
    """In-place pruning of a complete-array HostTree using validation rows.

    Returns the number of pruned internal nodes. Xv is the encoded [F, Nv]
    validation matrix; categorical routing uses tree.masks."""
    total = tree.feat.size
    depth = tree.max_depth
    nv = Xv.shape[1]
    stats = np.zeros((total, 3), dtype=np.float64)  # count, sum, sumsq
    cur = np.zeros(nv, dtype=np.int64)
    for _level in range(depth + 1):
        np.add.at(stats[:, 0], cur, 1.0)
        np.add.at(stats[:, 1], cur, yv)
        np.add.at(stats[:, 2], cur, yv * yv)
        f = tree.feat[cur]
        live = f >= 0
        if not live.any():
            break
        idx = np.nonzero(live)[0]
        fi = f[idx]
        xv = Xv[fi, idx]
        right = np.zeros(len(idx), dtype=np.int64)
        is_cat = cat_feats[fi] if cat_feats is not None else \
            np.zeros(len(idx), dtype=bool)
        ni = ~is_cat
        if ni.any():
            thr = bnd[fi[ni], tree.bin[cur[idx][ni]]]
            right[ni] = (xv[ni] > thr).astype(np.int64)
        if is_cat.any() and tree.masks is not None:
            ci = np.nonzero(is_cat)[0]
            codes = np.clip(xv[ci].astype(np.int64), 0, 255)
            node_sel = cur[idx][ci]
            words = tree.masks[node_sel, codes >> 6]
            right[ci] = ((words >> np.uint64(1) *
                          (codes & 63).astype(np.uint64)) &
                         np.uint64(1)).astype(np.int64)
        cur[idx] = 2 * cur[idx] + 1 + right

    # validation loss if node n becomes a leaf predicting tree.leaf_value[n]
    v = tree.leaf_value.astype(np.float64)
    if task == Task.CLASSIFICATION:
        # leaf predicts class 1 iff v > 0.5; loss = misclassified count
        pred1 = v > 0.5
        loss_leaf = np.where(pred1, stats[:, 0] - stats[:, 1], stats[:, 1])
    else:
        loss_leaf = stats[:, 2] - 2 * v * stats[:, 1] + stats[:, 0] * v * v
    subtree = loss_leaf.copy()
    pruned = 0
    for level in range(depth - 1, -1, -1):
        base = (1 << level) - 1
        for rel in range(1 << level):
            n = base + rel
            if tree.feat[n] < 0:
                continue
            child_loss = subtree[2 * n + 1] + subtree[2 * n + 2]
            if loss_leaf[n] <= child_loss + 1e-12:
                tree.feat[n] = -1
                pruned += 1
            else:
                subtree[n] = child_loss
    return pruned
