"""Tree inspection (capability analogue of ydf.model.tree: Python-side
tree structure access, port/python/ydf/model/tree/)."""
from __future__ import annotations

import dataclasses
from typing import Optional

from ydf_amd.model.forest import FlatForest


@dataclasses.dataclass
class Leaf:
    value: float
    cover: float = 0.0  # number of (weighted) training examples


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
    na_pos: bool = False  # missing values follow the positive child
    cover: float = 0.0
    # categorical-SET condition (vocab can exceed 256): token codes whose
    # presence sends the example to the positive child, or None
    set_items: Optional[tuple] = None


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
        cov = float(forest.cover[n]) if forest.cover is not None else 0.0
        if f < 0:
            return Leaf(value=float(forest.thr[n]), cover=cov)
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
        st = None
        if forest.set_idx is not None:
            si = int(forest.set_idx[n])
            if si >= 0:
                s0 = int(forest.set_offs[si])
                s1 = int(forest.set_offs[si + 1])
                st = tuple(int(v) for v in forest.set_items[s0:s1])
        return NonLeaf(feature=f, threshold=float(forest.thr[n]),
                       neg_child=build(left), pos_child=build(left + 1),
                       mask=mask, oblique=obl,
                       na_pos=bool(forest.na_right[n]), cover=cov,
                       set_items=st)

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


class TreePlot:
    """Self-contained SVG tree rendering (capability analogue of
    ydf.model plot_tree, port/python/ydf/model/tree/plot.py +
    plotter.js — that one is d3-based; this one emits static SVG so it
    renders anywhere, including offline notebooks)."""

    def __init__(self, svg: str):
        self._svg = svg

    def _repr_html_(self) -> str:
        return self._svg

    def html(self) -> str:
        return self._svg

    def to_file(self, path: str) -> None:
        with open(path, "w") as f:
            f.write(self._svg)


def plot_tree(tree: Tree, dataspec=None, max_depth: int = 6,
              label_classes=None) -> TreePlot:
    names = None
    if dataspec is not None:
        names = [c.name for c in dataspec.feature_columns]

    ROW_H, COL_W, BOX_W, BOX_H = 46, 190, 168, 34

    # layout: leaves get consecutive rows; internal nodes center over
    # their children. Depth-pruned subtrees render as "..." stubs.
    pos = {}
    next_row = [0]

    def layout(n, depth):
        if isinstance(n, Leaf) or depth >= max_depth:
            r = next_row[0]
            next_row[0] += 1
            pos[id(n)] = (depth, r)
            return r
        r0 = layout(n.neg_child, depth + 1)
        r1 = layout(n.pos_child, depth + 1)
        r = (r0 + r1) / 2.0
        pos[id(n)] = (depth, r)
        return r

    layout(tree.root, 0)
    width = (min(max_depth, tree.depth()) + 1) * COL_W + 40
    height = next_row[0] * ROW_H + 30

    def esc(s):
        return (str(s).replace("&", "&amp;").replace("<", "&lt;")
                .replace(">", "&gt;"))

    def cond_text(n):
        fname = names[n.feature] if names else f"f{n.feature}"
        if n.oblique is not None:
            return f"Σ wᵢ·xᵢ > {n.threshold:.4g}"
        if n.mask is not None:
            k = sum(bin(int(w)).count("1") for w in n.mask)
            return f"{fname} ∈ {{{k} values}}"
        if n.set_items is not None:
            return f"{fname} ∩ {{{len(n.set_items)} tokens}}"
        return f"{fname} > {n.threshold:.4g}"

    parts = []

    def draw(n, depth):
        d, r = pos[id(n)]
        x = 20 + d * COL_W
        y = 15 + r * ROW_H
        pruned = not isinstance(n, Leaf) and depth >= max_depth
        if isinstance(n, Leaf):
            txt, fill = f"value={n.value:.5g}", "#e8f4e8"
        elif pruned:
            txt, fill = "…", "#eeeeee"
        else:
            txt, fill = cond_text(n), "#e8eef8"
        sub = (f"n={n.cover:.0f}" if getattr(n, "cover", 0.0) else "")
        parts.append(
            f'<rect x="{x}" y="{y}" width="{BOX_W}" height="{BOX_H}" '
            f'rx="5" fill="{fill}" stroke="#667"/>'
            f'<text x="{x + 6}" y="{y + 15}" font-size="11" '
            f'font-family="monospace">{esc(txt)}</text>'
            + (f'<text x="{x + 6}" y="{y + 28}" font-size="9" '
               f'fill="#556" font-family="monospace">{esc(sub)}</text>'
               if sub else ""))
        if isinstance(n, Leaf) or pruned:
            return
        for child, lab in ((n.pos_child, "yes"), (n.neg_child, "no")):
            cd, cr = pos[id(child)]
            x2 = 20 + cd * COL_W
            y2 = 15 + cr * ROW_H + BOX_H / 2
            parts.append(
                f'<path d="M{x + BOX_W},{y + BOX_H / 2} C'
                f'{x + BOX_W + 40},{y + BOX_H / 2} {x2 - 40},{y2} '
                f'{x2},{y2}" fill="none" stroke="#99a"/>'
                f'<text x="{x + BOX_W + 8}" '
                f'y="{(y + BOX_H / 2 + y2) / 2 - 3}" font-size="9" '
                f'fill="#778">{lab}</text>')
            draw(child, depth + 1)

    draw(tree.root, 0)
    svg = (f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
           f'height="{height}" viewBox="0 0 {width} {height}">'
           + "".join(parts) + "</svg>")
    return TreePlot(svg)


def build_forest_from_trees(trees, n_features: int):
    """Inverse of extract_tree (capability analogue of the reference
    TreeBuilder, model/decision_tree/builder.h): assembles a FlatForest
    from Python Leaf/NonLeaf trees (numerical + categorical-mask +
    oblique conditions)."""
    import numpy as np

    feats, thrs, lefts, covers, cidx = [], [], [], [], []
    masks = []
    obl_ranges, obl_attr, obl_w = [], [], []
    roots, na_pos = [], []
    set_idx, set_offs, set_items = [], [0], []

    def new_slot():
        feats.append(-1)
        thrs.append(0.0)
        lefts.append(0)
        cidx.append(-1)
        covers.append(0.0)
        na_pos.append(0)
        set_idx.append(-1)
        return len(feats) - 1

    def fill(node, slot):
        covers[slot] = float(getattr(node, "cover", 0.0))
        if isinstance(node, Leaf):
            thrs[slot] = float(node.value)
            return
        if node.feature < 0 or node.feature >= n_features:
            raise ValueError(f"feature index {node.feature} out of range")
        feats[slot] = int(node.feature)
        na_pos[slot] = 1 if getattr(node, "na_pos", False) else 0
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
        elif getattr(node, "set_items", None) is not None:
            set_idx[slot] = len(set_offs) - 1
            set_items.extend(int(v) for v in node.set_items)
            set_offs.append(len(set_items))
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
        cover=np.asarray(covers, np.float32),
        obl_ranges=np.asarray(obl_ranges, np.int32).reshape(-1, 2),
        obl_attr=np.asarray(obl_attr, np.int32),
        obl_w=np.asarray(obl_w, np.float32),
        na_right=np.asarray(na_pos, np.uint8),
        set_idx=np.asarray(set_idx, np.int32) if set_items else None,
        set_offs=np.asarray(set_offs, np.int64) if set_items else None,
        set_items=np.asarray(set_items, np.int32) if set_items else None)


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
