"""Imports reference Yggdrasil Decision Forests models (read-only).

Decodes the reference's on-disk model directory (model/model_library.cc:
92-107: header.pb + data_spec.pb + <prefix>gradient_boosted_trees_header.pb /
random_forest_header.pb + nodes-XXXXX-of-YYYYY + done) with a hand-rolled
protobuf *wire-format* reader — no protoc, no schema files; the field
numbers below are documented against the reference .proto sources.

Supported: GBT (binary/multiclass/regression) and RF models with numerical
(Higher/DiscretizedHigher), categorical (ContainsBitmap/ContainsVector) and
boolean (TrueValue) and oblique (sparse linear projection) conditions.
Vector-sequence conditions are not supported (ROADMAP).
"""
from __future__ import annotations

import gzip
import os
import struct
from typing import Dict, Iterator, List, Optional, Tuple

import numpy as np

from ydf_amd.dataset.dataspec import (ColumnSpec, DataSpecification,
                                      OOV_ITEM, Semantic, Task)
from ydf_amd.model.forest import FlatForest
from ydf_amd.model.specialized import (GradientBoostedTreesModel,
                                       RandomForestModel)


# ---------------------------------------------------------------------------
# Minimal protobuf wire-format reader
# ---------------------------------------------------------------------------
class Wire:
    def __init__(self, data: bytes):
        self.d = data
        self.p = 0
        self.n = len(data)

    def varint(self) -> int:
        r = 0
        s = 0
        while True:
            b = self.d[self.p]
            self.p += 1
            r |= (b & 0x7F) << s
            if not b & 0x80:
                return r
            s += 7

    def fields(self) -> Iterator[Tuple[int, int, object]]:
        """Yields (field_number, wire_type, value)."""
        while self.p < self.n:
            tag = self.varint()
            fn, wt = tag >> 3, tag & 7
            if wt == 0:
                yield fn, wt, self.varint()
            elif wt == 1:
                v = struct.unpack_from("<d", self.d, self.p)[0]
                self.p += 8
                yield fn, wt, v
            elif wt == 2:
                ln = self.varint()
                yield fn, wt, self.d[self.p:self.p + ln]
                self.p += ln
            elif wt == 5:
                v = struct.unpack_from("<f", self.d, self.p)[0]
                self.p += 4
                yield fn, wt, v
            else:
                raise ValueError(f"unsupported wire type {wt}")


def _msg(data: bytes) -> Dict[int, list]:
    out: Dict[int, list] = {}
    for fn, _, v in Wire(data).fields():
        out.setdefault(fn, []).append(v)
    return out


def _f32(v) -> float:
    if isinstance(v, float):
        return v
    return struct.unpack("<f", struct.pack("<I", v & 0xFFFFFFFF))[0]


def _packed_varints(b: bytes) -> List[int]:
    w = Wire(b)
    out = []
    while w.p < w.n:
        out.append(w.varint())
    return out


def _packed_floats(b: bytes) -> List[float]:
    return list(struct.unpack(f"<{len(b) // 4}f", b))


def _packed_doubles(b: bytes) -> List[float]:
    return list(struct.unpack(f"<{len(b) // 8}d", b))


# ---------------------------------------------------------------------------
# Blob sequence (reference utils/blob_sequence.h:125-150): "BS" magic,
# u16 version, u8 compression, 3 reserved bytes; then records of
# u32-LE length + payload.
# ---------------------------------------------------------------------------
def read_blob_sequence(path: str) -> Iterator[bytes]:
    with open(path, "rb") as f:
        data = f.read()
    if data[:2] != b"BS":
        raise ValueError(f"{path}: not a blob sequence")
    version, compression = struct.unpack_from("<HB", data, 2)
    p = 8
    if compression == 1:
        data = data[:8] + gzip.decompress(data[8:])
    while p + 4 <= len(data):
        (ln,) = struct.unpack_from("<I", data, p)
        p += 4
        yield data[p:p + ln]
        p += ln


# ---------------------------------------------------------------------------
# data_spec.pb (reference dataset/data_spec.proto)
# ---------------------------------------------------------------------------
_COLTYPE = {1: Semantic.NUMERICAL, 4: Semantic.CATEGORICAL,
            7: Semantic.BOOLEAN, 9: Semantic.DISCRETIZED_NUMERICAL,
            10: Semantic.HASH, 5: Semantic.CATEGORICAL_SET,
            11: Semantic.NUMERICAL_VECTOR_SEQUENCE}


def parse_data_spec(raw: bytes):
    """Returns (columns, boundaries_by_col). Column fields: type=1, name=2,
    numerical=5 {mean=1,min=2,max=3}, categorical=6 {number_of_unique=2,
    is_already_integerized=5, items map=7 {key=1,value=2{index=1,count=2}}},
    discretized_numerical=8 {boundaries=1 packed f32}, boolean=9."""
    spec = _msg(raw)
    columns = []
    disc_bounds = {}
    for ci, colb in enumerate(spec.get(1, [])):
        col = _msg(colb)
        ctype = col.get(1, [1])[0]
        name = col.get(2, [b""])[0].decode()
        sem = _COLTYPE.get(ctype, Semantic.NUMERICAL)
        mean = 0.0
        vocab = None
        if 5 in col:  # numerical spec
            num = _msg(col[5][0])
            mean = _f32(num.get(1, [0.0])[0]) if 1 in num else 0.0
        if sem in (Semantic.CATEGORICAL, Semantic.CATEGORICAL_SET) \
                and 6 in col:
            cat = _msg(col[6][0])
            n_unique = cat.get(2, [0])[0]
            integerized = bool(cat.get(5, [0])[0])
            if integerized:
                vocab = [str(i) for i in range(n_unique)]
                vocab[0] = OOV_ITEM
            else:
                vocab = [OOV_ITEM] * max(n_unique, 1)
                for entry in cat.get(7, []):
                    e = _msg(entry)
                    key = e.get(1, [b""])[0].decode()
                    val = _msg(e.get(2, [b""])[0])
                    idx = val.get(1, [0])[0]
                    if idx < len(vocab):
                        vocab[idx] = key
        if sem == Semantic.DISCRETIZED_NUMERICAL and 8 in col:
            dn = _msg(col[8][0])
            if 1 in dn:
                disc_bounds[ci] = np.asarray(_packed_floats(dn[1][0]),
                                             dtype=np.float32)
        vec_dim = 0
        if sem == Semantic.NUMERICAL_VECTOR_SEQUENCE and 13 in col:
            vs = _msg(col[13][0])
            vec_dim = int(vs.get(1, [0])[0])
        columns.append(ColumnSpec(
            name=name,
            semantic=(Semantic.NUMERICAL
                      if sem == Semantic.DISCRETIZED_NUMERICAL else sem),
            vocab=vocab, mean=mean, vecseq_dim=vec_dim))
    return columns, disc_bounds


# ---------------------------------------------------------------------------
# Node records (reference model/decision_tree/decision_tree.proto): Node
# {classifier=1 {top_value=1, distribution=2 {counts=1 packed double,
# sum=2}}, regressor=2 {top_value=1}, condition=3 NodeCondition
# {na_value=1, attribute=2, condition=3 Condition{higher=2{threshold=1},
# true_value=3, contains=4{elements=1 packed}, contains_bitmap=5{bytes=1},
# discretized_higher=6{threshold=1}}},
# num_training_examples_with_weight=5}.
# Written pre-order, NEGATIVE child first (decision_tree.cc:580-585).
# ---------------------------------------------------------------------------
class _NodeRec:
    __slots__ = ("is_leaf", "attr", "thr", "mask", "value", "cover",
                 "na_value", "obl", "elements", "bitmap", "vecseq")


def parse_node(raw: bytes, disc_bounds, n_classes: int,
               binary_class_idx: int = 2) -> _NodeRec:
    node = _msg(raw)
    r = _NodeRec()
    r.is_leaf = 3 not in node
    r.attr = -1
    r.thr = 0.0
    r.mask = None
    r.value = 0.0
    r.cover = 0.0
    r.na_value = False
    r.obl = None
    r.elements = None
    r.bitmap = None
    r.vecseq = None
    if 5 in node:
        pass
    if 1 in node:  # classifier output
        cls = _msg(node[1][0])
        if 2 in cls:
            dist = _msg(cls[2][0])
            counts = _packed_doubles(dist.get(1, [b""])[0]) \
                if dist.get(1) else []
            total = dist.get(2, [0.0])[0]
            if n_classes == 2:
                c = counts[binary_class_idx] if len(counts) > \
                    binary_class_idx else 0.0
                r.value = c / total if total else 0.0
            else:
                r.value = [c / total if total else 0.0
                           for c in counts[1:1 + n_classes]]
        else:
            r.value = float(cls.get(1, [0])[0])
    elif 2 in node:  # regressor output
        reg = _msg(node[2][0])
        r.value = _f32(reg.get(1, [0.0])[0]) if 1 in reg else 0.0
    elif 5 in node:  # uplift leaf (NodeUpliftOutput)
        up = _msg(node[5][0])
        te = _packed_floats(up[4][0]) if up.get(4) else []
        if te:
            r.value = float(te[0])
        else:
            swpt = _packed_doubles(up[2][0]) if up.get(2) else []
            swpto = _packed_doubles(up[3][0]) if up.get(3) else []
            rt = swpto[1] / swpt[1] if len(swpt) > 1 and swpt[1] else 0.0
            rc = swpto[0] / swpt[0] if swpt and swpt[0] else 0.0
            r.value = rt - rc
    elif 6 in node:  # anomaly-detection leaf: example count; the
        # importer converts to depth + c(n) while walking the tree
        an = _msg(node[6][0])
        r.value = ("anomaly", int(an.get(1, [0])[0]))
    if not r.is_leaf:
        cond = _msg(node[3][0])
        r.na_value = bool(cond.get(1, [0])[0])
        r.attr = cond.get(2, [0])[0]
        inner = _msg(cond.get(3, [b""])[0])
        if 4 in cond:
            r.cover = float(cond[4][0])
        if 2 in inner:  # Higher: value >= threshold
            thr = _msg(inner[2][0])
            t = _f32(thr.get(1, [0.0])[0]) if 1 in thr else 0.0
            # our kernels test strict >; x >= t  <=>  x > nextafter(t, -inf)
            r.thr = float(np.nextafter(np.float32(t), np.float32("-inf")))
        elif 6 in inner:  # DiscretizedHigher: bin_index >= threshold
            thr = _msg(inner[6][0])
            t = int(thr.get(1, [0])[0])
            b = disc_bounds.get(r.attr)
            if b is None or t - 1 >= len(b) or t < 1:
                raise ValueError("discretized condition without boundaries")
            # bin(v) >= t  <=>  v >= boundaries[t-1]
            r.thr = float(np.nextafter(np.float32(b[t - 1]),
                                       np.float32("-inf")))
        elif 3 in inner:  # TrueValue (boolean): v == True -> positive
            r.thr = 0.5
        elif 4 in inner:  # ContainsVector
            els = _msg(inner[4][0])
            r.elements = _packed_varints(els[1][0]) if els.get(1) else []
            m = np.zeros(4, dtype=np.uint64)
            for e in r.elements:
                if e < 256:
                    m[e >> 6] |= np.uint64(1 << (e & 63))
            r.mask = m
        elif 7 in inner:  # Oblique: sum_i w_i * x[a_i] >= threshold
            # (decision_tree.proto:114-131: attributes=1 packed,
            # weights=2 packed f32, threshold=3)
            ob = _msg(inner[7][0])
            attrs = _packed_varints(ob[1][0]) if ob.get(1) else []
            ws = _packed_floats(ob[2][0]) if ob.get(2) else []
            t = _f32(ob.get(3, [0.0])[0]) if 3 in ob else 0.0
            r.obl = (attrs, ws)
            # our kernels test strict >; dot >= t  <=>  dot > nextafter down
            r.thr = float(np.nextafter(np.float32(t), np.float32("-inf")))
        elif 8 in inner:  # NumericalVectorSequence condition
            vs = _msg(inner[8][0])
            if 2 in vs:  # ProjectedMoreThan: exists <v|anchor> >= t
                pm = _msg(vs[2][0])
                anc = _msg(pm.get(1, [b""])[0])
                anchor = np.asarray(
                    _packed_floats(anc[1][0]) if anc.get(1) else [],
                    dtype=np.float32)
                t = _f32(pm.get(2, [0.0])[0]) if 2 in pm else 0.0
                # internal virtual column holds max_dot with strict >
                r.thr = float(np.nextafter(np.float32(t),
                                           np.float32("-inf")))
                r.vecseq = ("dot", anchor)
            elif 1 in vs:  # CloserThan: exists |v-anchor|^2 <= t2
                ct = _msg(vs[1][0])
                anc = _msg(ct.get(1, [b""])[0])
                anchor = np.asarray(
                    _packed_floats(anc[1][0]) if anc.get(1) else [],
                    dtype=np.float32)
                t2 = _f32(ct.get(2, [0.0])[0]) if 2 in ct else 0.0
                # internal column holds -min_sq with strict >
                r.thr = float(np.nextafter(np.float32(-t2),
                                           np.float32("-inf")))
                r.vecseq = ("dist", anchor)
            else:
                raise ValueError("empty vector-sequence condition")
        elif 5 in inner:  # ContainsBitmap
            bm = _msg(inner[5][0]).get(1, [b""])[0]
            r.bitmap = bm
            m = np.zeros(32, dtype=np.uint8)
            m[:min(len(bm), 32)] = np.frombuffer(bm[:32], dtype=np.uint8)
            r.mask = m.view(np.uint64)
        else:
            raise ValueError(
                f"unsupported condition type (fields {list(inner)})")
    return r


# ---------------------------------------------------------------------------
# Model assembly
# ---------------------------------------------------------------------------
_VECSEQ_BASE = 1 << 20  # sentinel feature ids for vecseq conditions


def _read_trees(model_dir: str, prefix: str, disc_bounds, n_classes,
                value_scale: float = 1.0, wta: bool = False,
                set_feats=frozenset(), vecseq_reg=None):
    shards = sorted(p for p in os.listdir(model_dir)
                    if p.startswith(prefix + "nodes-"))
    records: List[bytes] = []
    for s in shards:
        records.extend(read_blob_sequence(os.path.join(model_dir, s)))
    feats, thrs, lefts, roots, cidx, masks, covers = \
        [], [], [], [], [], [], []
    obl_ranges, obl_attr, obl_w = [], [], []
    na_right = []
    set_idx_l, set_items, set_offs = [], [], [0]
    pos = 0

    def new_slot():
        feats.append(-1)
        thrs.append(0.0)
        lefts.append(0)
        cidx.append(-1)
        covers.append(0.0)
        na_right.append(0)
        set_idx_l.append(-1)
        return len(feats) - 1

    def fill_node(idx, depth=0):
        # consumes the next record into slot idx; children are allocated as
        # an ADJACENT pair (our flat layout needs right == left + 1, which
        # the on-disk DFS pre-order does not give for free)
        nonlocal pos
        rec = parse_node(records[pos], disc_bounds, n_classes)
        pos += 1
        covers[idx] = rec.cover
        if rec.is_leaf:
            if isinstance(rec.value, tuple) and rec.value[0] == "anomaly":
                # isolation-forest leaf: score contribution is the path
                # length depth + c(n_leaf) (isolation_forest.h:51)
                from ydf_amd.model.specialized import IsolationForestModel

                thrs[idx] = depth + \
                    IsolationForestModel.expected_path_length(
                        float(rec.value[1]))
            elif wta and not isinstance(rec.value, list):
                # winner-take-all: the tree votes its majority class
                # (binary: leaf P(class2) > 0.5 -> vote 1)
                thrs[idx] = 1.0 if rec.value > 0.5 else 0.0
            else:
                thrs[idx] = float(rec.value) * value_scale \
                    if not isinstance(rec.value, list) else 0.0
            return
        feats[idx] = rec.attr
        na_right[idx] = 1 if rec.na_value else 0
        if rec.vecseq is not None and vecseq_reg is not None:
            kind, anchor = rec.vecseq
            key = (int(rec.attr), kind, anchor.tobytes())
            if key not in vecseq_reg:
                vecseq_reg[key] = (len(vecseq_reg), int(rec.attr), kind,
                                   anchor)
            feats[idx] = _VECSEQ_BASE + vecseq_reg[key][0]
            thrs[idx] = rec.thr
        elif rec.mask is not None and (
                rec.attr in set_feats
                or (rec.elements is not None and len(rec.elements)
                    and max(rec.elements) >= 256)):
            # categorical-SET condition, or a plain-categorical
            # ContainsVector whose dictionary exceeds the 256-bit mask
            # (large-vocab training): keep the full element list
            if rec.elements is not None:
                items = list(rec.elements)
            else:
                bits = np.unpackbits(
                    np.frombuffer(rec.bitmap, dtype=np.uint8),
                    bitorder="little")
                items = list(np.nonzero(bits)[0])
            set_idx_l[idx] = len(set_offs) - 1
            set_items.extend(int(v) for v in items)
            set_offs.append(len(set_items))
        elif rec.mask is not None:
            cidx[idx] = len(masks)
            masks.append(rec.mask)
        elif rec.obl is not None:
            attrs, ws = rec.obl
            feats[idx] = attrs[0] if attrs else 0
            cidx[idx] = -(2 + len(obl_ranges))
            obl_ranges.append((len(obl_attr), len(attrs)))
            obl_attr.extend(attrs)
            obl_w.extend(ws)
            thrs[idx] = rec.thr
        else:
            thrs[idx] = rec.thr
        li = new_slot()
        new_slot()
        lefts[idx] = li
        fill_node(li, depth + 1)    # negative child first on disk
        fill_node(li + 1, depth + 1)

    while pos < len(records):
        root = new_slot()
        roots.append(root)
        fill_node(root)
    return FlatForest(
        feat=np.asarray(feats, np.int32), thr=np.asarray(thrs, np.float32),
        left=np.asarray(lefts, np.int32), roots=np.asarray(roots, np.int32),
        cat_idx=np.asarray(cidx, np.int32),
        masks=np.stack(masks).astype(np.uint64) if masks
        else np.zeros((0, 4), np.uint64),
        cover=np.asarray(covers, np.float32),
        obl_ranges=np.asarray(obl_ranges, np.int32).reshape(-1, 2),
        obl_attr=np.asarray(obl_attr, np.int32),
        obl_w=np.asarray(obl_w, np.float32),
        na_right=np.asarray(na_right, np.uint8),
        set_idx=np.asarray(set_idx_l, np.int32),
        set_offs=np.asarray(set_offs, np.int64),
        set_items=np.asarray(set_items, np.int32))


def load_ydf_model(path: str, file_prefix: str = ""):
    """Loads a reference YDF model directory as a ydf_amd model."""
    with open(os.path.join(path, file_prefix + "header.pb"), "rb") as f:
        header = _msg(f.read())
    with open(os.path.join(path, file_prefix + "data_spec.pb"), "rb") as f:
        columns, disc_bounds = parse_data_spec(f.read())
    task = Task(header.get(2, [1])[0]) if header.get(2, [1])[0] in (
        1, 2, 3, 4, 5, 6) else Task.CLASSIFICATION
    label_idx = header.get(3, [len(columns) - 1])[0]
    if label_idx >= (1 << 62):  # negative varint (-1): no label column
        label_idx = -1
    input_features = []
    for v in header.get(5, []):
        if isinstance(v, bytes):
            input_features.extend(_packed_varints(v))
        else:
            input_features.append(v)
    has_label = 0 <= label_idx < len(columns)
    label_name = columns[label_idx].name if has_label else None
    # order dataspec features like the model's input_features
    feat_cols = [columns[i] for i in input_features]
    dataspec = DataSpecification(
        columns=feat_cols + ([columns[label_idx]] if has_label else []),
        label=label_name)
    # remap attribute indices (original column idx -> dense feature idx)
    remap = {ci: i for i, ci in enumerate(input_features)}
    # vecseq conditions materialize as virtual projection columns (one
    # per distinct (source, kind, anchor)); _read_trees fills this
    # registry with sentinel feature ids >= _VECSEQ_BASE
    vecseq_reg = {}

    def _finish(forest):
        if vecseq_reg:
            virtuals = []
            base = len(feat_cols)
            for key, (vid, attr, kind, anchor) in sorted(
                    vecseq_reg.items(), key=lambda kv: kv[1][0]):
                src_name = columns[attr].name
                virtuals.append(ColumnSpec(
                    name=f"{src_name}.{kind}.{vid}",
                    semantic=Semantic.NUMERICAL,
                    vecseq_source=src_name, vecseq_kind=kind,
                    vecseq_anchor=np.asarray(anchor, np.float32).copy()))
                remap[_VECSEQ_BASE + vid] = base + vid
            dataspec.columns = feat_cols + virtuals + (
                [columns[label_idx]] if has_label else [])
        return _remap_forest(forest, remap)

    set_feats = frozenset(
        i for i, c in enumerate(columns)
        if c.semantic == Semantic.CATEGORICAL_SET)
    gbt_hdr_path = os.path.join(
        path, file_prefix + "gradient_boosted_trees_header.pb")
    rf_hdr_path = os.path.join(path, file_prefix + "random_forest_header.pb")
    label_vocab = columns[label_idx].vocab if has_label else None
    classes = list(label_vocab[1:]) if label_vocab else None
    n_classes = len(classes) if classes else 2

    if os.path.exists(gbt_hdr_path):
        # gbt header (model/gradient_boosted_trees/gradient_boosted_trees.
        # proto): num_trees=2, loss=3, initial_predictions=4 (repeated f32),
        # num_trees_per_iter=5
        with open(gbt_hdr_path, "rb") as f:
            gh = _msg(f.read())
        inits = [(_f32(v) if not isinstance(v, bytes) else None)
                 for v in gh.get(4, [])]
        init_preds = []
        for v in gh.get(4, []):
            if isinstance(v, bytes):
                init_preds.extend(_packed_floats(v))
            else:
                init_preds.append(_f32(v))
        if not init_preds:
            init_preds = [0.0]
        loss_ref = gh.get(3, [0])[0]
        # Reference proto::Loss enum -> our internal trainer loss ids
        # (they agree on 1,2,3,7,8,9; focal=6->11, xe-ndcg=5->12,
        # cox=10->13, deprecated ndcg5=4->9).
        loss = {1: 1, 2: 2, 3: 3, 4: 9, 5: 12, 6: 11, 7: 7, 8: 8,
                9: 9, 10: 13}.get(int(loss_ref), int(loss_ref))
        forest = _read_trees(path, file_prefix, disc_bounds, n_classes,
                             set_feats=set_feats, vecseq_reg=vecseq_reg)
        ntpi = gh.get(5, [1])[0]
        activation = "identity"
        if task == Task.CLASSIFICATION:
            activation = "softmax" if ntpi > 1 else "sigmoid"
        elif loss == 7:  # POISSON: reference applies exp at predict
            activation = "exp"
        if loss == 0:  # DEFAULT: infer like the reference does
            loss = (3 if ntpi > 1 else 1) \
                if task == Task.CLASSIFICATION else 2
        model = GradientBoostedTreesModel(
            forest=_finish(forest), dataspec=dataspec,
            task=task, label_classes=classes, init_predictions=init_preds,
            num_trees_per_iter=ntpi, activation=activation,
            metadata={"imported_from": "yggdrasil-decision-forests",
                      "loss": int(loss)})
        return model
    if os.path.exists(rf_hdr_path):
        # rf header (model/random_forest/random_forest.proto): num_trees=2,
        # winner_take_all_inference=3 (default true)
        with open(rf_hdr_path, "rb") as f:
            rh = _msg(f.read())
        wta = bool(rh.get(3, [1])[0])
        forest = _read_trees(path, file_prefix, disc_bounds, n_classes,
                             wta=wta and task == Task.CLASSIFICATION,
                             set_feats=set_feats, vecseq_reg=vecseq_reg)
        meta = {"imported_from": "yggdrasil-decision-forests",
                "winner_take_all": wta}
        if task in (Task.CATEGORICAL_UPLIFT, Task.NUMERICAL_UPLIFT):
            # AbstractModel.uplift_treatment_col_idx = 9
            tcol = header.get(9, [-1])[0]
            if 0 <= tcol < len(columns):
                meta["uplift_treatment"] = columns[tcol].name
                tv = columns[tcol].vocab
                meta["treatment_vocab"] = list(tv[1:]) if tv else None
        model = RandomForestModel(
            forest=_finish(forest), dataspec=dataspec,
            task=task, label_classes=classes,
            init_predictions=[0.0],
            num_trees_per_iter=1, activation="identity",
            metadata=meta)
        return model
    if_hdr_path = os.path.join(path,
                               file_prefix + "isolation_forest_header.pb")
    if os.path.exists(if_hdr_path):
        # isolation_forest.proto: num_trees=2, num_examples_per_trees=4
        from ydf_amd.model.specialized import IsolationForestModel

        with open(if_hdr_path, "rb") as f:
            ih = _msg(f.read())
        forest = _read_trees(path, file_prefix, disc_bounds, n_classes,
                             vecseq_reg=vecseq_reg)
        return IsolationForestModel(
            forest=_finish(forest), dataspec=dataspec,
            task=Task.ANOMALY_DETECTION, init_predictions=[0.0],
            num_trees_per_iter=1, activation="identity",
            num_examples_per_tree=int(ih.get(4, [256])[0]),
            metadata={"imported_from": "yggdrasil-decision-forests"})
    raise ValueError(f"unsupported or missing model header in {path}")


def _remap_forest(forest: FlatForest, remap: Dict[int, int]) -> FlatForest:
    feat = forest.feat.copy()
    for i in range(len(feat)):
        if feat[i] >= 0:
            feat[i] = remap.get(int(feat[i]), 0)
    forest.feat = feat
    if len(forest.obl_attr):
        oa = forest.obl_attr.copy()
        for i in range(len(oa)):
            oa[i] = remap.get(int(oa[i]), 0)
        forest.obl_attr = oa
    return forest
