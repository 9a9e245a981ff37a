"""Tree inspection (capability analogue of ydf.model.tree: Python-side
tree structure access, port/python/ydf/model/tree/)."""
from __future__ import annotations

import dataclasses
from typing import Optional

from ydf_amd.model.forest import FlatForest


@dataclasses.dataclass
class Leaf:
    value: float


@dataclasses.dataclass
class NonLeaf:
    feature: int
    threshold: float
    neg_child: "Node"  # x <= threshold
    pos_child: "Node"  # x > threshold
    # categorical set-split: 256-bit "goes positive" mask (4 x u64), or None
    mask: Optional[tuple] = None
    # oblique: ((attr, ...), (weight, ...)) with sum w*x > threshold, or None
    oblique: Optional[tuple] = None


Node = object  # Leaf | NonLeaf


@dataclasses.dataclass
class Tree:
    root: Node

    def num_nodes(self) -> int:
        def count(n):
            if isinstance(n, Leaf):
                return 1
            return 1 + count(n.neg_child) + count(n.pos_child)

        return count(self.root)

    def depth(self) -> int:
        def d(n):
            if isinstance(n, Leaf):
                return 0
            return 1 + max(d(n.neg_child), d(n.pos_child))

        return d(self.root)


def extract_tree(forest: FlatForest, idx: int) -> Tree:
    lo, hi = forest.tree_slice(idx)

    def build(n: int):
        f = int(forest.feat[n])
        if f < 0:
            return Leaf(value=float(forest.thr[n]))
        left = int(forest.left[n])
        ci = int(forest.cat_idx[n])
        mask = tuple(int(w) for w in forest.masks[ci]) if ci >= 0 else None
        obl = None
        if ci <= -2:
            oi = -(ci + 2)
            s0, nn = int(forest.obl_ranges[oi, 0]), int(
                forest.obl_ranges[oi, 1])
            obl = (tuple(int(a) for a in forest.obl_attr[s0:s0 + nn]),
                   tuple(float(w) for w in forest.obl_w[s0:s0 + nn]))
        return NonLeaf(feature=f, threshold=float(forest.thr[n]),
                       neg_child=build(left), pos_child=build(left + 1),
                       mask=mask, oblique=obl)

    return Tree(root=build(int(forest.roots[idx])))


def format_tree(tree: Tree, dataspec=None, max_depth: int = 6) -> str:
    names = None
    if dataspec is not None:
        names = [c.name for c in dataspec.feature_columns]

    out = []

    def fmt(n, depth, prefix):
        if depth > max_depth:
            out.append(prefix + "...")
            return
        if isinstance(n, Leaf):
            out.append(prefix + f"value={n.value:.6g}")
            return
        fname = names[n.feature] if names else f"f{n.feature}"
        if n.oblique is not None:
            attrs, ws = n.oblique
            terms = " + ".join(
                f"{w:.4g}*{names[a] if names else f'f{a}'}"
                for a, w in zip(attrs, ws))
            out.append(prefix + f"{terms} > {n.threshold:.6g}")
        elif n.mask is not None:
            out.append(prefix + f"{fname!r} in mask[{n.mask[0]:#x},...]")
        else:
            out.append(prefix + f"{fname!r} > {n.threshold:.6g}")
        fmt(n.pos_child, depth + 1, prefix + "    ├(yes) ")
        fmt(n.neg_child, depth + 1, prefix + "    └(no)  ")

    fmt(tree.root, 0, "")
    return "\n".join(out)


def build_forest_from_trees(trees, n_features: int):
    """Inverse of extract_tree (capability analogue of the reference
    TreeBuilder, model/decision_tree/builder.h): assembles a FlatForest
    from Python Leaf/NonLeaf trees (numerical + categorical-mask +
    oblique conditions)."""
    import numpy as np

    feats, thrs, lefts, covers, cidx = [], [], [], [], []
    masks = []
    obl_ranges, obl_attr, obl_w = [], [], []
    roots = []

    def new_slot():
        feats.append(-1)
        thrs.append(0.0)
        lefts.append(0)
        cidx.append(-1)
        covers.append(0.0)
        return len(feats) - 1

    def fill(node, slot):
        if isinstance(node, Leaf):
            thrs[slot] = float(node.value)
            return
        if node.feature < 0 or node.feature >= n_features:
            raise ValueError(f"feature index {node.feature} out of range")
        feats[slot] = int(node.feature)
        if node.oblique is not None:
            attrs, ws = node.oblique
            cidx[slot] = -(2 + len(obl_ranges))
            obl_ranges.append((len(obl_attr), len(attrs)))
            obl_attr.extend(int(a) for a in attrs)
            obl_w.extend(float(w) for w in ws)
            thrs[slot] = float(node.threshold)
        elif node.mask is not None:
            cidx[slot] = len(masks)
            masks.append(np.asarray(node.mask, dtype=np.uint64))
            thrs[slot] = 0.0
        else:
            thrs[slot] = float(node.threshold)
        li = new_slot()
        new_slot()
        lefts[slot] = li
        fill(node.neg_child, li)
        fill(node.pos_child, li + 1)

    for t in trees:
        root = t.root if isinstance(t, Tree) else t
        slot = new_slot()
        roots.append(slot)
        fill(root, slot)
    import numpy as np

    return FlatForest(
        feat=np.asarray(feats, np.int32),
        thr=np.asarray(thrs, np.float32),
        left=np.asarray(lefts, np.int32),
        roots=np.asarray(roots, np.int32),
        cat_idx=np.asarray(cidx, np.int32),
        masks=np.stack(masks).astype(np.uint64) if masks
        else np.zeros((0, 4), np.uint64),
        obl_ranges=np.asarray(obl_ranges, np.int32).reshape(-1, 2),
        obl_attr=np.asarray(obl_attr, np.int32),
        obl_w=np.asarray(obl_w, np.float32))


def build_model_from_trees(trees, dataspec, task=None,
                           init_predictions=None, activation="identity",
                           model_type="GRADIENT_BOOSTED_TREES"):
    """Assembles a servable model from hand-built trees (mirrors
    constructing a model with the reference TreeBuilder + headers)."""
    from ydf_amd.dataset.dataspec import Task
    from ydf_amd.model.specialized import MODEL_CLASSES

    forest = build_forest_from_trees(
        trees, n_features=len(dataspec.feature_columns))
    cls = MODEL_CLASSES[model_type]
    return cls(forest=forest, dataspec=dataspec,
               task=task or Task.REGRESSION,
               init_predictions=init_predictions or [0.0],
               num_trees_per_iter=1, activation=activation,
               metadata={"hand_built": True})
