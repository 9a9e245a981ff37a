"""Exact (non-binned) numerical split training — the CPU oracle.

The reference trains EXACT numerical splits by default
(learner/decision_tree/splitter_scanner.h:1290 presorted scan;
preprocessing.h:106 PresortNumericalFeatures); the MI355X hot path
always trains on the 256-bin quantile representation (documented
deviation). This module supplies the reference-default semantics on
CPU: a presorted, level-wise, vectorized exact splitter used when
`discretize_numerical_columns=False` on device="cpu", and as the
correctness oracle that bounds the quality loss of binned training.

Design is numpy level-wise (all nodes of a level scanned in one pass
per feature via segment prefix sums) — an original formulation, not a
translation of the reference's per-node bucket scan. Gain / leaf-value
formulas match ydf_amd.ops split_scan exactly (hessian gain with
l1/l2), so exact-vs-binned differences isolate the effect of binning.

Thresholds use the reference's midpoint rule (utils.h:103 MidThreshold:
finite, > left value, equal to right value when no float fits between).
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np


def _l1_thresh(g: np.ndarray, l1: float) -> np.ndarray:
    if l1 <= 0.0:
        return g
    return np.sign(g) * np.maximum(np.abs(g) - l1, 0.0)


def _mid_threshold(a: np.ndarray, b: np.ndarray) -> np.ndarray:
    """(a+b)/2 in f32, guaranteed finite and > a (falls back to b)."""
    a = a.astype(np.float32)
    b = b.astype(np.float32)
    t = a + (b - a) / np.float32(2.0)
    return np.where(t <= a, b, t)


@dataclasses.dataclass
class ExactTree:
    feat: np.ndarray      # i32 [nodes], -1 leaf
    thr: np.ndarray       # f32 [nodes]: threshold (x > thr -> right) / leaf
    left: np.ndarray      # i32 [nodes], right = left + 1
    cover: np.ndarray     # f32 [nodes]
    cat_mask: dict        # node -> u64[4] "category goes right" bitmask
    gain: np.ndarray      # f32 [nodes]


class ExactSplitter:
    """Level-wise exact splitter over a raw feature-major f32 matrix."""

    def __init__(self, X: np.ndarray, cat_flags: Optional[np.ndarray],
                 max_depth: int = 6, min_examples: int = 5,
                 min_hessian: float = 1e-3, lambda_l2: float = 0.0,
                 lambda_l1: float = 0.0, cat_smooth: float = 1.0,
                 min_gain: float = 0.0):
        assert X.ndim == 2
        self.X = np.ascontiguousarray(X, dtype=np.float32)
        self.F, self.N = X.shape
        self.cat_flags = (np.zeros(self.F, dtype=bool)
                          if cat_flags is None
                          else np.asarray(cat_flags, dtype=bool))
        self.max_depth = max_depth
        self.min_examples = min_examples
        self.min_hessian = min_hessian
        self.lambda_l2 = lambda_l2
        self.lambda_l1 = lambda_l1
        self.cat_smooth = cat_smooth
        self.min_gain = min_gain
        # one global presort per numerical feature (reference
        # FORCE_PRESORT strategy); stable so ties keep row order
        self.order = {f: np.argsort(self.X[f], kind="stable")
                      for f in range(self.F) if not self.cat_flags[f]}
        self.n_cats = {
            f: int(self.X[f].max()) + 1 if self.N else 1
            for f in range(self.F) if self.cat_flags[f]}

    # -- one level, one feature: best split per node ---------------------
    def _scan_numerical(self, f: int, node_of_row: np.ndarray,
                        n_nodes: int, g: np.ndarray, h: np.ndarray):
        ord_f = self.order[f]
        nid = node_of_row[ord_f]
        live = nid >= 0
        seq_n = nid[live]
        if seq_n.size == 0:
            return None
        # stable counting-sort by node id -> per-node runs, each sorted
        # by feature value (global presort is preserved within a run)
        by_node = np.argsort(seq_n, kind="stable")
        sn = seq_n[by_node]
        rows = ord_f[live][by_node]
        sv = self.X[f][rows]
        sg = g[rows].astype(np.float64)
        sh = h[rows].astype(np.float64)
        # segment boundaries
        seg_start = np.zeros(n_nodes + 1, dtype=np.int64)
        np.add.at(seg_start, sn + 1, 1)
        seg_start = np.cumsum(seg_start)
        cg = np.cumsum(sg)
        ch = np.cumsum(sh)
        # candidate boundary after position i (left = [seg_start..i])
        pos = np.arange(sn.size)
        same_node = np.empty(sn.size, dtype=bool)
        same_node[:-1] = sn[:-1] == sn[1:]
        same_node[-1] = False
        distinct = np.empty(sn.size, dtype=bool)
        distinct[:-1] = sv[:-1] != sv[1:]
        distinct[-1] = False
        cand = same_node & distinct
        if not cand.any():
            return None
        base_g = np.concatenate([[0.0], cg])[seg_start[sn]]
        base_h = np.concatenate([[0.0], ch])[seg_start[sn]]
        GL = cg - base_g
        HL = ch - base_h
        CL = pos - seg_start[sn] + 1
        segG = (np.concatenate([[0.0], cg])[seg_start[1:]]
                - np.concatenate([[0.0], cg])[seg_start[:-1]])
        segH = (np.concatenate([[0.0], ch])[seg_start[1:]]
                - np.concatenate([[0.0], ch])[seg_start[:-1]])
        segC = seg_start[1:] - seg_start[:-1]
        GR = segG[sn] - GL
        HR = segH[sn] - HL
        CR = segC[sn] - CL
        ok = cand & (CL >= self.min_examples) & (CR >= self.min_examples) \
            & (HL >= self.min_hessian) & (HR >= self.min_hessian)
        tl = _l1_thresh(GL, self.lambda_l1)
        tr = _l1_thresh(GR, self.lambda_l1)
        tp = _l1_thresh(segG[sn], self.lambda_l1)
        with np.errstate(divide="ignore", invalid="ignore"):
            gain = tl * tl / (HL + self.lambda_l2) \
                + tr * tr / (HR + self.lambda_l2) \
                - tp * tp / (segH[sn] + self.lambda_l2)
        gain = np.where(ok & np.isfinite(gain), gain, -np.inf)
        # per-node argmax via reduceat over segments
        starts = seg_start[:-1]
        valid_seg = segC > 0
        best = np.full(n_nodes, -np.inf)
        best_pos = np.full(n_nodes, -1, dtype=np.int64)
        if valid_seg.any():
            red = np.maximum.reduceat(gain, np.maximum(starts[valid_seg],
                                                       0))
            best[valid_seg] = red
            # recover argmax positions
            for s_i, node in enumerate(np.nonzero(valid_seg)[0]):
                s0 = seg_start[node]
                s1 = seg_start[node + 1]
                if best[node] > -np.inf:
                    best_pos[node] = s0 + int(np.argmax(gain[s0:s1]))
        thr = np.zeros(n_nodes, dtype=np.float32)
        has = best_pos >= 0
        if has.any():
            p = best_pos[has]
            thr[has] = _mid_threshold(sv[p], sv[p + 1])
        return best, thr, None

    def _scan_categorical(self, f: int, node_of_row: np.ndarray,
                          n_nodes: int, g: np.ndarray, h: np.ndarray):
        K = self.n_cats[f]
        live = node_of_row >= 0
        nid = node_of_row[live]
        codes = self.X[f][live].astype(np.int64)
        key = nid * K + codes
        cnt = np.bincount(key, minlength=n_nodes * K).reshape(n_nodes, K)
        sumg = np.bincount(key, weights=g[live],
                           minlength=n_nodes * K).reshape(n_nodes, K)
        sumh = np.bincount(key, weights=h[live],
                           minlength=n_nodes * K).reshape(n_nodes, K)
        # CART one-vs-rest: order categories by the smoothed -g/h leaf
        # statistic, then scan like a numerical feature
        ratio = -sumg / (sumh + self.cat_smooth)
        order = np.argsort(ratio, axis=1, kind="stable")
        og = np.take_along_axis(sumg, order, axis=1)
        oh = np.take_along_axis(sumh, order, axis=1)
        oc = np.take_along_axis(cnt, order, axis=1)
        GL = np.cumsum(og, axis=1)[:, :-1]
        HL = np.cumsum(oh, axis=1)[:, :-1]
        CL = np.cumsum(oc, axis=1)[:, :-1]
        G = sumg.sum(axis=1, keepdims=True)
        H = sumh.sum(axis=1, keepdims=True)
        C = cnt.sum(axis=1, keepdims=True)
        GR, HR, CR = G - GL, H - HL, C - CL
        ok = (CL >= self.min_examples) & (CR >= self.min_examples) \
            & (HL >= self.min_hessian) & (HR >= self.min_hessian)
        tl = _l1_thresh(GL, self.lambda_l1)
        tr = _l1_thresh(GR, self.lambda_l1)
        tp = _l1_thresh(G, self.lambda_l1)
        with np.errstate(divide="ignore", invalid="ignore"):
            gain = tl * tl / (HL + self.lambda_l2) \
                + tr * tr / (HR + self.lambda_l2) \
                - tp * tp / (H + self.lambda_l2)
        gain = np.where(ok & np.isfinite(gain), gain, -np.inf)
        best_k = np.argmax(gain, axis=1)
        best = gain[np.arange(n_nodes), best_k]
        masks = np.zeros((n_nodes, 4), dtype=np.uint64)
        for n in range(n_nodes):
            if np.isfinite(best[n]):
                right_cats = order[n, best_k[n] + 1:]
                for c in right_cats:
                    if c < 256:
                        masks[n, c >> 6] |= np.uint64(1 << (int(c) & 63))
        return best, None, masks

    # -- grow one tree ---------------------------------------------------
    def grow_tree(self, g: np.ndarray, h: np.ndarray) -> ExactTree:
        N = self.N
        node_of_row = np.zeros(N, dtype=np.int64)
        feat: List[int] = [-1]
        thr: List[float] = [0.0]
        left: List[int] = [0]
        cover: List[float] = [float(N)]
        gains: List[float] = [0.0]
        cat_mask: dict = {}
        # leaf value of a node from its (G,H)
        stats = {0: (float(g.sum()), float(h.sum()), N)}
        frontier = [0]
        # split levels 0..max_depth-1 (same convention as the binned
        # trainer: max_depth=6 -> up to 64 leaves)
        for depth in range(self.max_depth):
            if not frontier:
                break
            # relabel frontier nodes to 0..K-1 for the scans
            relabel = {n: i for i, n in enumerate(frontier)}
            lab = np.full(N, -1, dtype=np.int64)
            for n, i in relabel.items():
                lab[node_of_row == n] = i
            K = len(frontier)
            best_gain = np.full(K, -np.inf)
            best_feat = np.full(K, -1, dtype=np.int64)
            best_thr = np.zeros(K, dtype=np.float32)
            best_mask = {}
            for f in range(self.F):
                if self.cat_flags[f]:
                    res = self._scan_categorical(f, lab, K, g, h)
                else:
                    res = self._scan_numerical(f, lab, K, g, h)
                if res is None:
                    continue
                bg, bt, bm = res
                upd = bg > best_gain
                for i in np.nonzero(upd)[0]:
                    best_gain[i] = bg[i]
                    best_feat[i] = f
                    if bt is not None:
                        best_thr[i] = bt[i]
                    if bm is not None:
                        best_mask[i] = bm[i]
            new_frontier = []
            for i, n in enumerate(frontier):
                if best_feat[i] < 0 or not np.isfinite(best_gain[i]) \
                        or best_gain[i] <= self.min_gain:
                    continue
                f = int(best_feat[i])
                li = len(feat)
                feat[n] = f
                left[n] = li
                gains[n] = float(best_gain[i])
                rows = node_of_row == n
                if self.cat_flags[f]:
                    m = best_mask[i]
                    cat_mask[n] = m
                    codes = self.X[f][rows].astype(np.int64)
                    shifted = np.right_shift(
                        m[codes >> 6], (codes & 63).astype(np.uint64))
                    go_right = (shifted & np.uint64(1)).astype(bool)
                    thr[n] = 0.0
                else:
                    t = best_thr[i]
                    thr[n] = float(t)
                    go_right = self.X[f][rows] > t
                ridx = np.nonzero(rows)[0]
                feat.extend([-1, -1])
                thr.extend([0.0, 0.0])
                left.extend([0, 0])
                gains.extend([0.0, 0.0])
                lrows = ridx[~go_right]
                rrows = ridx[go_right]
                node_of_row[lrows] = li
                node_of_row[rrows] = li + 1
                cover.extend([float(lrows.size), float(rrows.size)])
                stats[li] = (float(g[lrows].sum()), float(h[lrows].sum()),
                             lrows.size)
                stats[li + 1] = (float(g[rrows].sum()),
                                 float(h[rrows].sum()), rrows.size)
                new_frontier.extend([li, li + 1])
            frontier = new_frontier
        # leaf values
        for n in range(len(feat)):
            if feat[n] < 0 and n in stats:
                G, H, _ = stats[n]
                v = 0.0
                if H != 0.0:
                    tg = G
                    if self.lambda_l1 > 0:
                        tg = np.sign(G) * max(abs(G) - self.lambda_l1, 0.0)
                    v = -tg / (H + self.lambda_l2)
                thr[n] = float(v)
        self._last_node_of_row = node_of_row
        return ExactTree(feat=np.asarray(feat, dtype=np.int32),
                         thr=np.asarray(thr, dtype=np.float32),
                         left=np.asarray(left, dtype=np.int32),
                         cover=np.asarray(cover, dtype=np.float32),
                         cat_mask=cat_mask,
                         gain=np.asarray(gains, dtype=np.float32))


def exact_trees_to_forest(trees: List[ExactTree], shrinkage: float):
    """Concatenates exact trees into a FlatForest (leaf values scaled by
    shrinkage, matching build_flat_forest semantics)."""
    from ydf_amd.model.forest import FlatForest

    feat, thr, left, roots, cover = [], [], [], [], []
    cat_idx, masks = [], []
    base = 0
    for t in trees:
        roots.append(base)
        n = len(t.feat)
        feat.append(t.feat)
        tt = t.thr.copy()
        leaf = t.feat < 0
        tt[leaf] *= shrinkage
        thr.append(tt)
        left.append(np.where(t.feat >= 0, t.left + base, 0))
        cover.append(t.cover)
        ci = np.full(n, -1, dtype=np.int32)
        for node, m in t.cat_mask.items():
            ci[node] = len(masks)
            masks.append(m)
        cat_idx.append(ci)
        base += n
    return FlatForest(
        feat=np.concatenate(feat), thr=np.concatenate(thr),
        left=np.concatenate(left),
        roots=np.asarray(roots, dtype=np.int32),
        cat_idx=np.concatenate(cat_idx),
        masks=(np.asarray(masks, dtype=np.uint64).reshape(-1, 4)
               if masks else None),
        cover=np.concatenate(cover))


def train_gbt_exact(X: np.ndarray, y: np.ndarray,
                    cat_flags: Optional[np.ndarray], loss: int,
                    num_trees: int, shrinkage: float, max_depth: int,
                    min_examples: int, min_hessian: float,
                    lambda_l1: float, lambda_l2: float,
                    cat_smooth: float = 1.0, min_gain: float = 0.0):
    """Exact-split GBT boosting loop (binomial=1 / squared-error=2).

    Gradient formulas mirror ops.grad_hess (cpu_ops.cpp cpu_grad_hess)
    so exact-vs-binned differences isolate the binning. Returns
    (trees, init_prediction)."""
    N = X.shape[1]
    yf = np.asarray(y, dtype=np.float64)
    if loss == 1:  # binomial: init = log-odds of the base rate
        p = np.clip(yf.mean(), 1e-6, 1 - 1e-6)
        init = float(np.log(p / (1 - p)))
    else:
        init = float(yf.mean())
    preds = np.full(N, init, dtype=np.float64)
    sp = ExactSplitter(X, cat_flags, max_depth=max_depth,
                       min_examples=min_examples,
                       min_hessian=min_hessian, lambda_l2=lambda_l2,
                       lambda_l1=lambda_l1, cat_smooth=cat_smooth,
                       min_gain=min_gain)
    trees: List[ExactTree] = []
    for _ in range(num_trees):
        if loss == 1:
            pr = 1.0 / (1.0 + np.exp(-preds))
            g = pr - yf
            h = np.maximum(pr * (1.0 - pr), 1e-16)
        else:
            g = preds - yf
            h = np.ones(N, dtype=np.float64)
        t = sp.grow_tree(g, h)
        trees.append(t)
        leaf_of_row = sp._last_node_of_row
        preds += shrinkage * t.thr[leaf_of_row].astype(np.float64)
    return trees, init


class MhldSplitter(ExactSplitter):
    """MHLD oblique splits (reference oblique.h:33-38 + oblique.cc
    FindBestConditionMHLDObliqueTemplate): greedily grow a feature
    subset (<= max_attributes); for each candidate subset the
    projection coefficients come from Linear Discriminant Analysis on
    the node's examples (w = Sw^-1 (mu1 - mu0) for the binary case),
    and the threshold from the regular numerical scan over the
    projected values. Classification only, like the reference
    (oblique.cc:690)."""

    def __init__(self, X, cat_flags, y01, max_attributes: int = 4,
                 **kw):
        super().__init__(X, cat_flags, **kw)
        self.y01 = np.asarray(y01, dtype=bool)
        self.max_attributes = max_attributes
        self.num_feats = [f for f in range(self.F)
                          if not self.cat_flags[f]]

    def _lda_direction(self, rows, feats):
        Xs = self.X[np.ix_(feats, rows)].astype(np.float64)  # [k, n]
        y = self.y01[rows]
        if y.all() or (~y).any() == 0 or not y.any():
            return None
        mu0 = Xs[:, ~y].mean(axis=1)
        mu1 = Xs[:, y].mean(axis=1)
        d0 = Xs[:, ~y] - mu0[:, None]
        d1 = Xs[:, y] - mu1[:, None]
        sw = d0 @ d0.T + d1 @ d1.T
        sw[np.diag_indices_from(sw)] += 1e-6 * max(1.0, np.trace(sw))
        try:
            w = np.linalg.solve(sw, mu1 - mu0)
        except np.linalg.LinAlgError:
            return None
        nrm = np.linalg.norm(w)
        if not np.isfinite(nrm) or nrm == 0:
            return None
        return w / nrm

    def _best_threshold(self, proj, g, h):
        """Best (gain, threshold) of a 1-D exact scan (sorted values)."""
        o = np.argsort(proj, kind="stable")
        sv = proj[o]
        cg = np.cumsum(g[o])
        ch = np.cumsum(h[o])
        G, H = cg[-1], ch[-1]
        n = len(o)
        pos = np.arange(1, n)
        valid = sv[:-1] != sv[1:]
        CL = pos
        CR = n - pos
        GL, HL = cg[:-1], ch[:-1]
        GR, HR = G - GL, H - HL
        ok = valid & (CL >= self.min_examples) & \
            (CR >= self.min_examples) & (HL >= self.min_hessian) & \
            (HR >= self.min_hessian)
        if not ok.any():
            return -np.inf, 0.0
        with np.errstate(divide="ignore", invalid="ignore"):
            gain = GL * GL / (HL + self.lambda_l2) \
                + GR * GR / (HR + self.lambda_l2) \
                - G * G / (H + self.lambda_l2)
        gain = np.where(ok & np.isfinite(gain), gain, -np.inf)
        i = int(np.argmax(gain))
        return float(gain[i]), float(_mid_threshold(
            np.float32(sv[i]), np.float32(sv[i + 1])))

    def grow_tree(self, g, h) -> ExactTree:
        """Greedy MHLD growth per node (recursive; node sizes shrink
        geometrically so the per-node LDA solves stay cheap)."""
        feat: list = [-1]
        thr: list = [0.0]
        left: list = [0]
        cover: list = [float(self.N)]
        gains: list = [0.0]
        oblique: dict = {}
        node_of_row = np.zeros(self.N, dtype=np.int64)

        def split_node(node, rows, depth):
            if depth >= self.max_depth or \
                    rows.size < 2 * self.min_examples:
                return
            # greedy subset growth
            subset: list = []
            best = (-np.inf, None, None)  # gain, w, thr
            improved = True
            while improved and len(subset) < self.max_attributes:
                improved = False
                for f in self.num_feats:
                    if f in subset:
                        continue
                    cand = subset + [f]
                    w = self._lda_direction(rows, cand)
                    if w is None:
                        continue
                    proj = (w[None, :] @ self.X[np.ix_(
                        cand, rows)].astype(np.float64))[0]
                    gain, t = self._best_threshold(
                        proj.astype(np.float32), g[rows], h[rows])
                    if gain > best[0]:
                        best = (gain, (list(cand), w.copy()), t)
                        chosen = f
                        improved = True
                if improved:
                    subset.append(chosen)
            if best[1] is None or best[0] <= self.min_gain:
                return
            attrs, w = best[1]
            t = best[2]
            proj = (w[None, :] @ self.X[np.ix_(
                attrs, rows)].astype(np.float64))[0]
            go_right = proj > t
            if go_right.all() or not go_right.any():
                return
            li = len(feat)
            feat[node] = attrs[0]
            left[node] = li
            thr[node] = t
            gains[node] = float(best[0])
            oblique[node] = (np.asarray(attrs, dtype=np.int32),
                             w.astype(np.float32), float(t))
            for _ in range(2):
                feat.append(-1)
                thr.append(0.0)
                left.append(0)
                gains.append(0.0)
                cover.append(0.0)
            lrows = rows[~go_right]
            rrows = rows[go_right]
            node_of_row[lrows] = li
            node_of_row[rrows] = li + 1
            cover[li] = float(lrows.size)
            cover[li + 1] = float(rrows.size)
            split_node(li, lrows, depth + 1)
            split_node(li + 1, rrows, depth + 1)

        split_node(0, np.arange(self.N), 0)
        # leaf values
        for n in range(len(feat)):
            if feat[n] < 0:
                rows = np.nonzero(node_of_row == n)[0] if n else None
                if n == 0 and len(feat) == 1:
                    rows = np.arange(self.N)
                if rows is None or rows.size == 0:
                    continue
                G = g[rows].sum()
                H = h[rows].sum()
                if H != 0.0:
                    thr[n] = float(-G / (H + self.lambda_l2))
        self._last_node_of_row = node_of_row
        t = ExactTree(feat=np.asarray(feat, dtype=np.int32),
                      thr=np.asarray(thr, dtype=np.float32),
                      left=np.asarray(left, dtype=np.int32),
                      cover=np.asarray(cover, dtype=np.float32),
                      cat_mask={}, gain=np.asarray(gains,
                                                   dtype=np.float32))
        t.oblique = oblique
        return t


def mhld_trees_to_forest(trees, shrinkage: float):
    """ExactTrees with .oblique records -> FlatForest with oblique
    conditions (cat_idx <= -2 indexing obl_* arrays)."""
    from ydf_amd.model.forest import FlatForest

    feat, thr, left, roots, cover, cat_idx = [], [], [], [], [], []
    obl_ranges, obl_attr, obl_w = [], [], []
    base = 0
    for t in trees:
        roots.append(base)
        n = len(t.feat)
        tt = t.thr.copy()
        leaf = t.feat < 0
        tt[leaf] *= shrinkage
        ci = np.full(n, -1, dtype=np.int32)
        for node, (attrs, w, thr_v) in t.oblique.items():
            ci[node] = -2 - len(obl_ranges)
            obl_ranges.append((len(obl_attr), len(attrs)))
            obl_attr.extend(int(a) for a in attrs)
            obl_w.extend(float(v) for v in w)
        feat.append(t.feat)
        thr.append(tt)
        left.append(np.where(t.feat >= 0, t.left + base, 0))
        cover.append(t.cover)
        cat_idx.append(ci)
        base += n
    return FlatForest(
        feat=np.concatenate(feat), thr=np.concatenate(thr),
        left=np.concatenate(left),
        roots=np.asarray(roots, dtype=np.int32),
        cat_idx=np.concatenate(cat_idx),
        cover=np.concatenate(cover),
        obl_ranges=np.asarray(obl_ranges, dtype=np.int32).reshape(-1, 2)
        if obl_ranges else None,
        obl_attr=np.asarray(obl_attr, dtype=np.int32)
        if obl_attr else None,
        obl_w=np.asarray(obl_w, dtype=np.float32) if obl_w else None)
