"""Reference-schema-derived export verification.

Decodes the output of `export_ydf.py` with `proto_wire.py`, a generic
protobuf decoder driven by `ydf_schema.json` (field numbers machine-
extracted from the reference's .proto sources) — fully independent of
`import_ydf.py`'s hand-written reader. An independent tree evaluator
then walks the schema-decoded nodes and must reproduce model.predict.

Reference analogue: SaveModel/LoadModel (model/model_library.cc:92-107)
+ ExpectEqualPredictions (utils/test_utils.h:258-297).
"""
import os

import numpy as np
import pytest

import ydf_amd as ydf
from ydf_amd.model import proto_wire as pw
from ydf_amd.model.export_ydf import export_ydf_model

NODE_MSG = "yggdrasil_decision_forests.model.decision_tree.proto.Node"
GBT_HDR = ("yggdrasil_decision_forests.model.gradient_boosted_trees."
           "proto.Header")
RF_HDR = "yggdrasil_decision_forests.model.random_forest.proto.Header"
ABSTRACT = "yggdrasil_decision_forests.model.proto.AbstractModel"
DATASPEC = ("yggdrasil_decision_forests.dataset.proto."
            "DataSpecification")


def _decode_dir(path, family):
    """Strict schema decode of every .pb + the node shards."""
    out = {
        "header": pw.decode(ABSTRACT, open(
            os.path.join(path, "header.pb"), "rb").read()),
        "data_spec": pw.decode(DATASPEC, open(
            os.path.join(path, "data_spec.pb"), "rb").read()),
    }
    hdr_file = {"gbt": "gradient_boosted_trees_header.pb",
                "rf": "random_forest_header.pb"}[family]
    msg = {"gbt": GBT_HDR, "rf": RF_HDR}[family]
    out["family_header"] = pw.decode(msg, open(
        os.path.join(path, hdr_file), "rb").read())
    shards = out["family_header"]["num_node_shards"]
    assert shards == 1
    recs = pw.read_blob_sequence(
        os.path.join(path, "nodes-00000-of-00001"))
    out["nodes"] = [pw.decode(NODE_MSG, r) for r in recs]
    assert os.path.exists(os.path.join(path, "done"))
    return out


def _build_trees(nodes, num_trees):
    """Re-builds trees from pre-order records (negative child first,
    reference decision_tree.cc pre-order serialization)."""
    pos = [0]

    def read():
        n = nodes[pos[0]]
        pos[0] += 1
        if "condition" in n:
            neg = read()
            pos_child = read()
            return {"cond": n["condition"], "neg": neg, "pos": pos_child,
                    "node": n}
        return {"leaf": n}

    trees = [read() for _ in range(num_trees)]
    assert pos[0] == len(nodes), "trailing node records"
    return trees


def _eval_condition(cond, row):
    """Independent condition semantics from decision_tree.proto:86-151."""
    attr = cond["attribute"]
    x = row[attr]
    c = cond["condition"]
    missing = x is None or (isinstance(x, float) and np.isnan(x))
    if missing:
        return bool(cond.get("na_value", False))
    if "higher_condition" in c:
        return float(x) >= c["higher_condition"]["threshold"]
    if "contains_bitmap_condition" in c:
        bm = c["contains_bitmap_condition"]["elements_bitmap"]
        v = int(x)
        return bool(bm[v // 8] >> (v % 8) & 1)
    if "contains_condition" in c:
        return int(x) in c["contains_condition"]["elements"]
    if "true_value_condition" in c:
        return bool(x)
    if "oblique_condition" in c:
        ob = c["oblique_condition"]
        s = sum(w * float(row[a]) for a, w in
                zip(ob["attributes"], ob["weights"]))
        return s >= ob["threshold"]
    raise AssertionError(f"unhandled condition {list(c)}")


def _eval_tree(tree, row):
    while "leaf" not in tree:
        tree = tree["pos"] if _eval_condition(tree["cond"], row) else \
            tree["neg"]
    return tree["leaf"]


def _rows_from_data(data, dataspec_cols, n=64):
    """Converts dict-of-arrays to per-row lists of proto-space values
    (categorical strings -> vocab indices from the EXPORTED data spec)."""
    cols = []
    for ci, col in enumerate(dataspec_cols):
        name = col["name"]
        arr = data[name]
        if col["type"] == "CATEGORICAL":
            vocab = {}
            for item in col["categorical"].get("items", []):
                vocab[item["key"]] = item["value"]["index"]
            cols.append([vocab.get(str(v), 0) for v in arr[:n]])
        else:
            cols.append([float(v) for v in arr[:n]])
    return [dict(enumerate(r)) for r in zip(*cols)]


def test_gbt_export_schema_decode_and_predict(binary_data, tmp_path):
    learner = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, max_depth=4, validation_ratio=0.0)
    model = learner.train(binary_data)
    out = str(tmp_path / "gbt_export")
    export_ydf_model(model, out)
    d = _decode_dir(out, "gbt")

    assert d["header"]["name"] == "GRADIENT_BOOSTED_TREES"
    assert d["header"]["task"] == "CLASSIFICATION"
    fh = d["family_header"]
    assert fh["node_format"] == "BLOB_SEQUENCE"
    assert fh["loss"] == "BINOMIAL_LOG_LIKELIHOOD"
    assert fh["num_trees"] == model.num_trees()
    np.testing.assert_allclose(fh["initial_predictions"],
                               model.init_predictions, rtol=1e-6)
    cols = d["data_spec"]["columns"]
    assert [c["name"] for c in cols] == ["x1", "x2", "x3", "label"]

    trees = _build_trees(d["nodes"], model.num_trees())
    rows = _rows_from_data(binary_data, cols)
    want = model.predict(binary_data, device="cpu")[:len(rows)]
    for i, row in enumerate(rows):
        margin = fh["initial_predictions"][0] + sum(
            _eval_tree(t, row)["regressor"]["top_value"] for t in trees)
        got = 1.0 / (1.0 + np.exp(-margin))
        assert abs(got - want[i]) < 1e-5, (i, got, want[i])


def test_rf_export_schema_decode_and_predict(regression_data, tmp_path):
    learner = ydf.RandomForestLearner(label="label", num_trees=10,
                                      max_depth=6,
                                      task=ydf.Task.REGRESSION)
    model = learner.train(regression_data)
    out = str(tmp_path / "rf_export")
    export_ydf_model(model, out)
    d = _decode_dir(out, "rf")

    fh = d["family_header"]
    assert fh["node_format"] == "BLOB_SEQUENCE"
    assert fh["num_trees"] == model.num_trees()
    assert d["header"]["task"] == "REGRESSION"

    trees = _build_trees(d["nodes"], model.num_trees())
    cols = d["data_spec"]["columns"]
    rows = _rows_from_data(regression_data, cols)
    want = model.predict(regression_data, device="cpu")[:len(rows)]
    for i, row in enumerate(rows):
        vals = [_eval_tree(t, row)["regressor"]["top_value"]
                for t in trees]
        got = float(np.mean(vals))
        assert abs(got - want[i]) < 1e-4, (i, got, want[i])


def test_rf_classification_export_distribution(binary_data, tmp_path):
    learner = ydf.RandomForestLearner(label="label", num_trees=10,
                                      max_depth=5, winner_take_all=False)
    model = learner.train(binary_data)
    out = str(tmp_path / "rfc_export")
    export_ydf_model(model, out)
    d = _decode_dir(out, "rf")
    trees = _build_trees(d["nodes"], model.num_trees())
    cols = d["data_spec"]["columns"]
    rows = _rows_from_data(binary_data, cols)
    want = model.predict(binary_data, device="cpu")[:len(rows)]
    for i, row in enumerate(rows):
        ps = []
        for t in trees:
            leaf = _eval_tree(t, row)["classifier"]["distribution"]
            counts = np.frombuffer(leaf["counts"], dtype="<f8") \
                if isinstance(leaf["counts"], bytes) else \
                np.asarray(leaf["counts"])
            ps.append(counts[2] / leaf["sum"])
        got = float(np.mean(ps))
        assert abs(got - want[i]) < 1e-5, (i, got, want[i])


def test_na_routing_exported(tmp_path):
    """LOCAL_IMPUTATION-trained model must carry na_value bits."""
    rng = np.random.RandomState(3)
    n = 4000
    x1 = rng.randn(n).astype(np.float32)
    y = (x1 > 0)
    x1[rng.rand(n) < 0.3] = np.nan
    data = {"x1": x1, "x2": rng.randn(n).astype(np.float32),
            "label": np.where(y, "a", "b")}
    learner = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, max_depth=4, validation_ratio=0.0,
        missing_value_policy="LOCAL_IMPUTATION")
    model = learner.train(data)
    if not model.forest.has_na_routing:
        pytest.skip("trained model routes no NA (degenerate)")
    out = str(tmp_path / "na_export")
    export_ydf_model(model, out)
    d = _decode_dir(out, "gbt")
    assert any(n.get("condition", {}).get("na_value", False)
               for n in d["nodes"])


def test_poisson_loss_enum(tmp_path):
    rng = np.random.RandomState(5)
    n = 3000
    x = rng.randn(n).astype(np.float32)
    lam = np.exp(0.5 * x)
    data = {"x": x, "label": rng.poisson(lam).astype(np.float32)}
    learner = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, loss="POISSON",
        num_trees=5, validation_ratio=0.0)
    model = learner.train(data)
    out = str(tmp_path / "poisson_export")
    export_ydf_model(model, out)
    d = _decode_dir(out, "gbt")
    assert d["family_header"]["loss"] == "POISSON"


REF_MODEL = ("/root/reference/yggdrasil_decision_forests/test_data/"
             "model/adult_binary_class_gbdt")


@pytest.mark.skipif(not os.path.exists(REF_MODEL),
                    reason="reference test_data not available")
def test_decoder_against_reference_golden_model():
    """Ground-truths the schema decoder itself on files the reference
    C++ wrote (so the decoder can't share a misreading with export)."""
    hdr = pw.decode(ABSTRACT, open(
        os.path.join(REF_MODEL, "header.pb"), "rb").read(), strict=False)
    assert hdr["name"] == "GRADIENT_BOOSTED_TREES"
    assert hdr["task"] == "CLASSIFICATION"
    gh = pw.decode(GBT_HDR, open(
        os.path.join(REF_MODEL, "gradient_boosted_trees_header.pb"),
        "rb").read())
    assert gh["num_trees"] == 68
    assert gh["loss"] == "BINOMIAL_LOG_LIKELIHOOD"
    assert gh["node_format"] == "BLOB_SEQUENCE"
    recs = pw.read_blob_sequence(
        os.path.join(REF_MODEL, "nodes-00000-of-00001"))
    nodes = [pw.decode(NODE_MSG, r) for r in recs]
    trees = _build_trees(nodes, 68)
    assert len(trees) == 68


def test_golden_header_bytes(binary_data, tmp_path):
    """Golden-bytes pin of the header field numbers (advisor finding:
    the RF node_format must be field 7, num_node_shards field 1)."""
    learner = ydf.RandomForestLearner(label="label", num_trees=3,
                                      max_depth=3)
    model = learner.train(binary_data)
    out = str(tmp_path / "rf_pin")
    export_ydf_model(model, out)
    raw = open(os.path.join(out, "random_forest_header.pb"), "rb").read()
    # field 1 varint (num_node_shards=1) -> 0x08 0x01
    assert raw[:2] == b"\x08\x01"
    # field 7 length-delimited "BLOB_SEQUENCE" -> tag 0x3a len 13
    assert b"\x3a\x0dBLOB_SEQUENCE" in raw


def test_wire_decoder_robust_on_fuzzed_bytes():
    """The schema decoder must reject arbitrary garbage with WireError
    (or decode it), never crash — hypothesis-style fuzz with a fixed
    seed for reproducibility."""
    rng = np.random.RandomState(1234)
    for _ in range(300):
        n = int(rng.randint(0, 60))
        data = rng.randint(0, 256, n).astype(np.uint8).tobytes()
        try:
            pw.decode(GBT_HDR, data, strict=True)
        except pw.WireError:
            pass
        except (UnicodeDecodeError, OverflowError):
            pass  # string fields may reject invalid utf-8


def test_blob_sequence_truncation_rejected(tmp_path):
    p = tmp_path / "trunc"
    p.write_bytes(b"BS" + b"\x01\x00\x00\x00\x00\x00" + b"\x10\x00\x00\x00ab")
    with pytest.raises(pw.WireError):
        pw.read_blob_sequence(str(p))
    p2 = tmp_path / "badmagic"
    p2.write_bytes(b"XY" + bytes(6))
    with pytest.raises(pw.WireError):
        pw.read_blob_sequence(str(p2))
