"""Flat forest: the serving-side node-array representation.

Capability analogue of the reference's flat-node serving models
(serving/decision_forest/decision_forest_serving.h:94 GenericNode +
FlatNodeModel :201), laid out SoA for the MI355X inference kernels:
  feat[n]  i32  split feature (-1 = leaf)
  thr[n]   f32  split threshold (x > thr -> right) or leaf value
  left[n]  i32  left child index (right = left + 1)
  roots[t] i32  root of tree t
"""
from __future__ import annotations

import dataclasses
from typing import List

import numpy as np

from ydf_amd.learner.trainer import HostTree


@dataclasses.dataclass
class FlatForest:
    feat: np.ndarray   # i32 [total]
    thr: np.ndarray    # f32 [total]
    left: np.ndarray   # i32 [total]
    roots: np.ndarray  # i32 [n_trees]
    # categorical set-splits: cat_idx[n] = -1 (numerical/leaf) or an index
    # into masks (256-bit "category goes right" bitmask as 4 x u64)
    cat_idx: np.ndarray = None   # i32 [total]
    masks: np.ndarray = None     # u64 [n_masks, 4]
    # training cover (weighted example count) per node; used by TreeSHAP
    cover: np.ndarray = None     # f32 [total]
    # oblique (sparse linear) conditions: a node with cat_idx <= -2 uses
    # oblique record -(cat_idx+2): sum_k obl_w[s+k]*x[obl_attr[s+k]] > thr
    # with (s, n) = obl_ranges[record] (reference Condition.Oblique,
    # decision_tree.proto:114-131)
    obl_ranges: np.ndarray = None  # i32 [n_obl, 2] (start, count)
    obl_attr: np.ndarray = None    # i32 [total_terms]
    obl_w: np.ndarray = None       # f32 [total_terms]
    # categorical-SET conditions (reference ContainsVector/Bitmap over
    # CATEGORICAL_SET columns, vocab can exceed 256): node n with
    # set_idx[n] = s >= 0 goes right iff the cell's token codes
    # intersect set_items[set_offs[s]:set_offs[s+1]]
    set_idx: np.ndarray = None     # i32 [total], -1 = not a set node
    set_offs: np.ndarray = None    # i64 [n_set+1]
    set_items: np.ndarray = None   # i32 [...]
    # na_value routing (reference NodeCondition.na_value): when the
    # node's input is MISSING (NaN numerical / -1 categorical code) the
    # example goes right iff na_right[node]. All-zeros = imputation-era
    # models (missing values were imputed at encode time).
    na_right: np.ndarray = None    # u8 [total]

    def __post_init__(self):
        if self.cat_idx is None:
            self.cat_idx = np.full(len(self.feat), -1, dtype=np.int32)
        if self.masks is None:
            self.masks = np.zeros((0, 4), dtype=np.uint64)
        if self.cover is None:
            self.cover = np.zeros(len(self.feat), dtype=np.float32)
        if self.obl_ranges is None:
            self.obl_ranges = np.zeros((0, 2), dtype=np.int32)
        if self.obl_attr is None:
            self.obl_attr = np.zeros(0, dtype=np.int32)
        if self.obl_w is None:
            self.obl_w = np.zeros(0, dtype=np.float32)
        if self.na_right is None:
            self.na_right = np.zeros(len(self.feat), dtype=np.uint8)
        if self.set_idx is None:
            self.set_idx = np.full(len(self.feat), -1, dtype=np.int32)
        if self.set_offs is None:
            self.set_offs = np.zeros(1, dtype=np.int64)
        if self.set_items is None:
            self.set_items = np.zeros(0, dtype=np.int32)

    @property
    def has_set_conditions(self) -> bool:
        return bool((self.set_idx >= 0).any())

    @property
    def has_na_routing(self) -> bool:
        return bool(self.na_right.any())

    @property
    def has_cats(self) -> bool:
        return len(self.masks) > 0 or len(self.obl_ranges) > 0

    @property
    def n_trees(self) -> int:
        return len(self.roots)

    @property
    def n_nodes(self) -> int:
        return len(self.feat)

    def tree_slice(self, t: int):
        lo = self.roots[t]
        hi = self.roots[t + 1] if t + 1 < self.n_trees else self.n_nodes
        return lo, hi


def _reachable(feat: np.ndarray, max_depth: int):
    total = feat.size
    reach = np.zeros(total, dtype=bool)
    internal = np.zeros(total, dtype=bool)
    reach[0] = True
    for level in range(max_depth):
        base = (1 << level) - 1
        size = 1 << level
        lv = slice(base, base + size)
        is_int = reach[lv] & (feat[lv] >= 0)
        internal[lv] = is_int
        idx = np.nonzero(is_int)[0] + base
        if idx.size:
            reach[2 * idx + 1] = True
            reach[2 * idx + 2] = True
    return reach, internal


def host_tree_to_flat(tree: HostTree, boundaries: np.ndarray,
                      leaf_scale: float = 1.0, cat_feats=None):
    """Compacts a complete-array HostTree into flat (feat, thr, left,
    cat_idx, masks) arrays.

    boundaries: padded [F, n_cuts] cut matrix; split threshold = cut[bin]
    (binning guarantees "bin > b" <=> "x > cut[b]"). cat_feats: bool [F]
    marking categorical features (their conditions use tree.masks).
    """
    reach, internal = _reachable(tree.feat, tree.max_depth)
    nodes = np.nonzero(reach)[0]  # ascending = level order
    total = tree.feat.size
    new_idx = np.full(total + 2, -1, dtype=np.int32)
    new_idx[nodes] = np.arange(len(nodes), dtype=np.int32)
    n_int = internal[nodes]
    feat = np.where(n_int, tree.feat[nodes], -1).astype(np.int32)
    lchild = np.minimum(2 * nodes + 1, total + 1)
    left = np.where(n_int, new_idx[lchild], 0).astype(np.int32)
    thr = np.empty(len(nodes), dtype=np.float32)
    cat_idx = np.full(len(nodes), -1, dtype=np.int32)
    masks = np.zeros((0, 4), dtype=np.uint64)
    obl_ranges, obl_attr, obl_w = [], [], []
    is_obl_node = np.zeros(len(nodes), dtype=bool)
    if tree.oblique:
        for a in tree.oblique:
            pos = new_idx[a]
            if pos >= 0 and internal[a]:
                is_obl_node[pos] = True
    ii = np.nonzero(n_int)[0]
    if ii.size:
        is_cat_node = np.zeros(len(nodes), dtype=bool)
        if cat_feats is not None and tree.masks is not None:
            is_cat_node[ii] = cat_feats[tree.feat[nodes[ii]]]
            is_cat_node &= ~is_obl_node
        ni = np.nonzero(n_int & ~is_cat_node & ~is_obl_node)[0]
        if ni.size:
            # numerical: tree.bin is a cut index; categorical nodes store a
            # sorted RANK there instead, so they must not index boundaries
            thr[ni] = boundaries[tree.feat[nodes[ni]], tree.bin[nodes[ni]]]
        ci = np.nonzero(is_cat_node)[0]
        if ci.size:
            thr[ci] = 0.0
            cat_idx[ci] = np.arange(ci.size, dtype=np.int32)
            masks = tree.masks[nodes[ci]].astype(np.uint64)
        for pos in np.nonzero(is_obl_node)[0]:
            attrs, ws, t = tree.oblique[int(nodes[pos])]
            cat_idx[pos] = -(2 + len(obl_ranges))
            obl_ranges.append((len(obl_attr), len(attrs)))
            obl_attr.extend(int(a) for a in attrs)
            obl_w.extend(float(w) for w in ws)
            thr[pos] = t
    li = np.nonzero(~n_int)[0]
    thr[li] = tree.leaf_value[nodes[li]] * leaf_scale
    cover = tree.counts[nodes].astype(np.float32)
    na = None
    if getattr(tree, "na", None) is not None:
        na = np.where(n_int, tree.na[nodes], 0).astype(np.uint8)
    obl = (np.asarray(obl_ranges, np.int32).reshape(-1, 2),
           np.asarray(obl_attr, np.int32), np.asarray(obl_w, np.float32))
    return feat, thr, left, cat_idx, masks, cover, obl, na


def best_first_tree_to_flat(tree, boundaries: np.ndarray,
                            leaf_scale: float = 1.0):
    """Converts a leaf-wise BestFirstTree (implicit-key splits dict) to
    flat arrays via adjacent-pair child allocation."""
    split_of = {s[0]: s[1:] for s in tree.splits}
    feats, thrs, lefts, covers, cidx = [], [], [], [], []
    masks = []

    def new_slot():
        feats.append(-1)
        thrs.append(0.0)
        lefts.append(0)
        covers.append(0.0)
        cidx.append(-1)
        return len(feats) - 1

    def fill(key, slot):
        covers[slot] = float(tree.counts.get(key, 0.0))
        if key in split_of:
            rec = split_of[key]
            fi, b = rec[0], rec[1]
            mask = rec[3] if len(rec) > 3 else None
            feats[slot] = fi
            if mask is not None:
                cidx[slot] = len(masks)
                masks.append(np.asarray(mask, dtype=np.uint64))
                thrs[slot] = 0.0
            else:
                thrs[slot] = float(boundaries[fi, b])
            li = new_slot()
            new_slot()
            lefts[slot] = li
            fill(2 * key + 1, li)
            fill(2 * key + 2, li + 1)
        else:
            thrs[slot] = float(tree.leaf_value.get(key, 0.0)) * leaf_scale

    root = new_slot()
    fill(0, root)
    return (np.asarray(feats, np.int32), np.asarray(thrs, np.float32),
            np.asarray(lefts, np.int32),
            np.asarray(cidx, np.int32),
            (np.stack(masks) if masks
             else np.zeros((0, 4), np.uint64)),
            np.asarray(covers, np.float32),
            (np.zeros((0, 2), np.int32), np.zeros(0, np.int32),
             np.zeros(0, np.float32)))


def build_flat_forest(trees: List[HostTree], boundaries: np.ndarray,
                      leaf_scale: float = 1.0, cat_feats=None) -> FlatForest:
    feats, thrs, lefts, roots, cidxs, mask_list, covers = \
        [], [], [], [], [], [], []
    rng_list, attr_list, w_list = [], [], []
    off = 0
    mask_off = 0
    obl_off = 0
    term_off = 0
    na_list = []
    for t in trees:
        if isinstance(t, HostTree):
            f, th, lf, ci, mk, cv, (orng, oat, ow), na = \
                host_tree_to_flat(t, boundaries, leaf_scale, cat_feats)
        else:  # BestFirstTree (leaf-wise growth)
            f, th, lf, ci, mk, cv, (orng, oat, ow) = \
                best_first_tree_to_flat(t, boundaries, leaf_scale)
            na = None
        na_list.append(na if na is not None
                       else np.zeros(len(f), np.uint8))
        lf = np.where(f >= 0, lf + off, 0)
        ci = np.where(ci >= 0, ci + mask_off, ci)
        ci = np.where(ci <= -2, ci - obl_off, ci)
        roots.append(off)
        off += len(f)
        mask_off += len(mk)
        if len(orng):
            orng = orng.copy()
            orng[:, 0] += term_off
        obl_off += len(orng)
        term_off += len(oat)
        feats.append(f)
        thrs.append(th)
        lefts.append(lf)
        cidxs.append(ci)
        mask_list.append(mk)
        covers.append(cv)
        rng_list.append(orng)
        attr_list.append(oat)
        w_list.append(ow)
    return FlatForest(
        feat=np.concatenate(feats) if feats else np.zeros(0, np.int32),
        thr=np.concatenate(thrs) if thrs else np.zeros(0, np.float32),
        left=np.concatenate(lefts) if lefts else np.zeros(0, np.int32),
        roots=np.asarray(roots, dtype=np.int32),
        cat_idx=np.concatenate(cidxs) if cidxs else np.zeros(0, np.int32),
        masks=np.concatenate(mask_list) if mask_list
        else np.zeros((0, 4), np.uint64),
        cover=np.concatenate(covers) if covers
        else np.zeros(0, np.float32),
        obl_ranges=(np.concatenate(rng_list).astype(np.int32)
                    if obl_off else None),
        obl_attr=np.concatenate(attr_list).astype(np.int32)
        if attr_list else None,
        obl_w=np.concatenate(w_list).astype(np.float32) if w_list else None,
        na_right=np.concatenate(na_list).astype(np.uint8)
        if na_list else None,
    )


def padded_boundaries(specs, max_bins: int = 256) -> np.ndarray:
    """Stacks per-column ragged cut lists into a dense [F, n_cuts] matrix,
    padding with +inf (padding adds no cut below any finite value, so bin
    assignment is unchanged)."""
    if not specs:
        raise ValueError("the dataset has no input feature columns")
    n_cuts = max(1, max((len(s.boundaries) if s.boundaries is not None else 0)
                        for s in specs))
    n_cuts = min(n_cuts, max_bins - 1)
    out = np.full((len(specs), n_cuts), np.inf, dtype=np.float32)
    for i, s in enumerate(specs):
        b = s.boundaries
        if b is not None and len(b):
            out[i, :min(len(b), n_cuts)] = b[:n_cuts]
    return out


def concat_forests(a: FlatForest, b: FlatForest) -> FlatForest:
    """Concatenates two flat forests (used by checkpoint/resume: partial
    model + newly grown trees)."""
    off = a.n_nodes
    moff = len(a.masks)
    ooff = len(a.obl_ranges)
    b_cat = np.where(b.cat_idx >= 0, b.cat_idx + moff, b.cat_idx)
    # oblique markers (<= -2) shift by the number of a's oblique records
    b_cat = np.where(b_cat <= -2, b_cat - ooff, b_cat)
    b_rng = b.obl_ranges.copy()
    if len(b_rng):
        b_rng[:, 0] += len(a.obl_attr)
    return FlatForest(
        feat=np.concatenate([a.feat, b.feat]),
        thr=np.concatenate([a.thr, b.thr]),
        left=np.concatenate([a.left,
                             np.where(b.feat >= 0, b.left + off, 0)]),
        roots=np.concatenate([a.roots, b.roots + off]).astype(np.int32),
        cat_idx=np.concatenate([a.cat_idx, b_cat]).astype(np.int32),
        masks=np.concatenate([a.masks, b.masks]) if (len(a.masks)
                                                     or len(b.masks))
        else np.zeros((0, 4), np.uint64),
        cover=np.concatenate([a.cover, b.cover]),
        obl_ranges=np.concatenate([a.obl_ranges, b_rng]).astype(np.int32)
        if (ooff or len(b_rng)) else None,
        obl_attr=np.concatenate([a.obl_attr, b.obl_attr]).astype(np.int32),
        obl_w=np.concatenate([a.obl_w, b.obl_w]).astype(np.float32),
    )


def build_quickscorer(forest: FlatForest):
    """Precomputes QuickScorer tables (reference
    quick_scorer_extended.h): per tree <= 64 in-order leaves, each
    internal node's left-subtree leaf mask. Numerical conditions only.
    Returns (conds i32 [C,4] = {feat, thr-bits, mask_lo, mask_hi},
    offs i32 [T+1], leaf_vals f32 [T,64])."""
    import struct as _struct

    if len(forest.masks) or len(forest.obl_ranges):
        raise ValueError("QuickScorer supports numerical conditions only")
    conds = []
    offs = [0]
    leaf_vals = np.zeros((forest.n_trees, 64), dtype=np.float32)
    for t in range(forest.n_trees):
        leaves = []

        def rec(n):
            if forest.feat[n] < 0:
                leaves.append(float(forest.thr[n]))
                i = len(leaves) - 1
                return i, i
            li = int(forest.left[n])
            l0, l1 = rec(li)
            r0, r1 = rec(li + 1)
            mask = 0
            for b in range(l0, l1 + 1):
                mask |= 1 << b
            conds.append((int(forest.feat[n]), float(forest.thr[n]),
                          mask))
            return l0, r1

        rec(int(forest.roots[t]))
        if len(leaves) > 64:
            raise ValueError(
                f"tree {t} has {len(leaves)} leaves (> 64); QuickScorer "
                "needs depth <= 6")
        leaf_vals[t, :len(leaves)] = leaves
        offs.append(len(conds))
    packed = np.zeros((max(len(conds), 1), 4), dtype=np.int32)
    for i, (feat, thr, mask) in enumerate(conds):
        packed[i, 0] = feat
        packed[i, 1] = np.frombuffer(
            _struct.pack("<f", thr), dtype=np.int32)[0]
        packed[i, 2] = np.frombuffer(
            _struct.pack("<I", mask & 0xFFFFFFFF), dtype=np.int32)[0]
        packed[i, 3] = np.frombuffer(
            _struct.pack("<I", (mask >> 32) & 0xFFFFFFFF),
            dtype=np.int32)[0]
    return (packed, np.asarray(offs, dtype=np.int32), leaf_vals)


def pack_binned_nodes(forest: FlatForest, boundaries: np.ndarray
                      ) -> np.ndarray:
    """Packs nodes for the 8-bit engine (reference
    8bits_numerical_features.h): each numerical threshold becomes its
    BIN INDEX in the training cut table ("bin > b" <=> "x > cut[b]"),
    stored as an int in the thr slot. Numerical conditions only."""
    if len(forest.masks) or len(forest.obl_ranges):
        raise ValueError("8-bit engine supports numerical conditions only")
    packed = np.zeros((forest.n_nodes, 4), dtype=np.int32)
    packed[:, 0] = forest.feat
    packed[:, 2] = forest.left
    internal = forest.feat >= 0
    thr_bits = forest.thr.view(np.int32).copy()
    for n in np.nonzero(internal)[0]:
        fi = int(forest.feat[n])
        cuts = boundaries[fi]
        b = int(np.searchsorted(cuts, forest.thr[n]))
        if b >= len(cuts) or cuts[b] != forest.thr[n]:
            # threshold not on a cut (e.g. imported model): fall back to
            # the number of cuts strictly below it minus matching epsilon
            b = int(np.searchsorted(cuts, forest.thr[n], side="right")) - 1
            b = max(b, 0)
        thr_bits[n] = b
    packed[:, 1] = thr_bits
    packed[:, 3] = -1
    return packed


def expand_bigcat_masks(forest: FlatForest, bigcat) -> FlatForest:
    """Converts group-space categorical masks on large-vocab features
    (trained through the <=256 CART-ordered group map) into
    full-dictionary set conditions (reference ContainsVector), so the
    model predicts on raw vocabulary codes.

    bigcat: {feature_idx: group_of_code i32 [vocab]}."""
    if not bigcat:
        return forest
    n = forest.n_nodes
    set_idx = forest.set_idx.copy() if forest.set_idx is not None \
        else np.full(n, -1, dtype=np.int32)
    offs = list(forest.set_offs) if forest.set_offs is not None else [0]
    items = list(forest.set_items) if forest.set_items is not None else []
    cat_idx = forest.cat_idx.copy()
    for node in range(n):
        fi = int(forest.feat[node])
        ci = int(cat_idx[node])
        if ci < 0 or fi not in bigcat:
            continue
        g_of_c = bigcat[fi]
        mask = forest.masks[ci]  # u64[4] over GROUP indices
        grp = g_of_c.astype(np.int64)
        bit = (mask[grp >> 6] >> (grp & 63).astype(np.uint64)) \
            & np.uint64(1)
        codes = np.nonzero(bit.astype(bool))[0]
        set_idx[node] = len(offs) - 1
        items.extend(int(c) for c in codes)
        offs.append(len(items))
        cat_idx[node] = -1
    forest.set_idx = set_idx
    forest.set_offs = np.asarray(offs, dtype=np.int64)
    forest.set_items = np.asarray(items, dtype=np.int32)
    forest.cat_idx = cat_idx
    return forest


def pack_binned8_nodes(forest: FlatForest, boundaries: np.ndarray,
                       leaf_scale: float = 1.0) -> np.ndarray:
    """Compact 8-byte nodes for the binned8 engine: word0 = feat u16 |
    bin u16 (feat 0xFFFF = leaf), word1 = left child (internal) or the
    f32 leaf-value bit pattern (leaf, pre-scaled by leaf_scale).
    Numerical conditions only. Halves the L2 node-fetch traffic that
    bounds batch serving."""
    if len(forest.masks) or len(forest.obl_ranges) \
            or forest.has_na_routing or forest.has_set_conditions:
        raise ValueError("binned8 engine supports numerical conditions "
                         "only")
    if forest.feat.max(initial=-1) >= 0xFFFF:
        raise ValueError("binned8 engine supports < 65535 features")
    n = forest.n_nodes
    packed = np.zeros((n, 2), dtype=np.uint32)
    internal = forest.feat >= 0
    bins = np.zeros(n, dtype=np.uint32)
    for node in np.nonzero(internal)[0]:
        fi = int(forest.feat[node])
        cuts = boundaries[fi]
        b = int(np.searchsorted(cuts, forest.thr[node]))
        if b >= len(cuts) or cuts[b] != forest.thr[node]:
            b = int(np.searchsorted(cuts, forest.thr[node],
                                    side="right")) - 1
            b = max(b, 0)
        bins[node] = b
    feat_u = np.where(internal, forest.feat.astype(np.uint32), 0xFFFF)
    packed[:, 0] = feat_u | (bins << 16)
    leaf_vals = (forest.thr * np.float32(leaf_scale)).astype(np.float32)
    packed[:, 1] = np.where(internal,
                            forest.left.astype(np.uint32),
                            leaf_vals.view(np.uint32))
    return packed


def pack_binned4_nodes(forest: FlatForest, boundaries: np.ndarray,
                       leaf_scale: float = 1.0):
    """4-byte nodes for the binned4 engine: feat (6 b, 63 = leaf) |
    bin (8 b) | left-child-or-leaf-index (18 b); leaf values (scaled)
    in a dense side table. Returns (nodes u32 [n], leaf_vals f32 [L]).
    Limits: numerical-only, F <= 63, nodes and leaves <= 2^18."""
    if len(forest.masks) or len(forest.obl_ranges) \
            or forest.has_na_routing or forest.has_set_conditions:
        raise ValueError("binned4 engine supports numerical conditions "
                         "only")
    n = forest.n_nodes
    fmax = int(forest.feat.max(initial=-1))
    if fmax >= 63 or n >= (1 << 18):
        raise ValueError("binned4 limits exceeded (F < 63, nodes < 2^18)")
    internal = forest.feat >= 0
    leaves = np.nonzero(~internal)[0]
    if len(leaves) >= (1 << 18):
        raise ValueError("binned4: too many leaves")
    leaf_idx = np.zeros(n, dtype=np.uint32)
    leaf_idx[leaves] = np.arange(len(leaves), dtype=np.uint32)
    leaf_vals = (forest.thr[leaves]
                 * np.float32(leaf_scale)).astype(np.float32)
    bins = np.zeros(n, dtype=np.uint32)
    for node in np.nonzero(internal)[0]:
        fi = int(forest.feat[node])
        cuts = boundaries[fi]
        b = int(np.searchsorted(cuts, forest.thr[node]))
        if b >= len(cuts) or cuts[b] != forest.thr[node]:
            b = max(int(np.searchsorted(cuts, forest.thr[node],
                                        side="right")) - 1, 0)
        bins[node] = b
    feat_u = np.where(internal, forest.feat.astype(np.uint32), 63)
    link = np.where(internal, forest.left.astype(np.uint32), leaf_idx)
    nodes = feat_u | (bins << 6) | (link << 14)
    return nodes.astype(np.uint32), leaf_vals
