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
