"""Golden-model import tests: load reference YDF models and reproduce the
reference's own prediction files (reference analogue: ExpectEqualPredictions
golden checks, utils/test_utils.h:258-297)."""
import os
import sys

import numpy as np
import pytest

import ydf_amd as ydf

BASE = "/root/reference/yggdrasil_decision_forests/test_data"

pytestmark = pytest.mark.skipif(not os.path.exists(BASE),
                                reason="reference test_data not available")
sys.setrecursionlimit(100000)


def _check(model_name, data_csv, pred_csv, col, tol):
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/{model_name}")
    te = pd.read_csv(f"{BASE}/dataset/{data_csv}")
    p = m.predict(te, device="cpu")
    g = pd.read_csv(f"{BASE}/prediction/{pred_csv}")[col].values
    assert np.abs(p - g).max() < tol, np.abs(p - g).max()
    return m


def test_golden_gbt_adult():
    m = _check("adult_binary_class_gbdt", "adult_test.csv",
               "adult_test_binary_class_gbdt.csv", ">50K", 1e-5)
    assert m.num_trees() == 68


def test_golden_gbt_abalone_regression():
    _check("abalone_regression_gbdt", "abalone.csv",
           "abalone_regression_gbdt.csv", "Rings", 1e-3)


def test_rf_small_models_load():
    for name in ("adult_binary_class_rf_wta_small",
                 "adult_binary_class_rf_nwta_small"):
        m = ydf.load_ydf_model(f"{BASE}/model/{name}")
        assert m.num_trees() == 10
        pd = pytest.importorskip("pandas")
        p = m.predict(pd.read_csv(f"{BASE}/dataset/adult_test.csv"),
                      device="cpu")
        assert 0.0 <= p.min() and p.max() <= 1.0
        assert 0.15 < p.mean() < 0.35  # base rate ~0.24


def test_multiclass_gbt_loads():
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/iris_multi_class_gbdt")
    p = m.predict(pd.read_csv(f"{BASE}/dataset/iris.csv"), device="cpu")
    assert p.shape[1] == 3
    np.testing.assert_allclose(p.sum(axis=1), 1.0, atol=1e-4)
    # the model should classify its own training data well
    labels = pd.read_csv(f"{BASE}/dataset/iris.csv")["class"].values
    pred_cls = np.asarray(m.label_classes)[p.argmax(axis=1)]
    assert (pred_cls == labels).mean() > 0.95


def test_imported_model_reserialization(tmp_path):
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/adult_binary_class_gbdt")
    p1 = m.predict(pd.read_csv(f"{BASE}/dataset/adult_test.csv"),
                   device="cpu")
    m.save(str(tmp_path / "m"))
    m2 = ydf.load_model(str(tmp_path / "m"))
    p2 = m2.predict(pd.read_csv(f"{BASE}/dataset/adult_test.csv"),
                    device="cpu")
    np.testing.assert_allclose(p1, p2, atol=1e-6)


def test_export_roundtrip(tmp_path, binary_data):
    """Our GBT model written in the reference format and read back through
    the importer must predict identically."""
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=15).train(
        binary_data)
    p = str(tmp_path / "ydf_format")
    ydf.export_ydf_model(m, p)
    assert os.path.exists(os.path.join(p, "done"))
    m2 = ydf.load_ydf_model(p)
    np.testing.assert_allclose(m.predict(binary_data, device="cpu"),
                               m2.predict(binary_data, device="cpu"),
                               atol=2e-6)


def test_export_roundtrip_categorical(tmp_path):
    rng = np.random.RandomState(3)
    n = 4000
    cats = rng.randint(0, 8, n)
    d = {"c": np.array([f"v{v}" for v in cats]),
         "x": rng.randn(n).astype(np.float32),
         "label": np.where((cats % 3 == 0) ^ (rng.rand(n) < 0.1),
                           "p", "n")}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=10,
                                        validation_ratio=0).train(d)
    p = str(tmp_path / "ydf_cat")
    ydf.export_ydf_model(m, p)
    m2 = ydf.load_ydf_model(p)
    np.testing.assert_allclose(m.predict(d, device="cpu"),
                               m2.predict(d, device="cpu"), atol=2e-6)


def test_oblique_gbt_import_quality(tmp_path):
    """Oblique (sparse linear projection) conditions: the reference's
    adult_binary_class_gbdt_oblique model must load and keep its quality
    (no golden prediction file exists for this model)."""
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/adult_binary_class_gbdt_oblique")
    f = m.forest
    assert (f.cat_idx <= -2).sum() > 0
    assert len(f.obl_ranges) == (f.cat_idx <= -2).sum()
    te = pd.read_csv(f"{BASE}/dataset/adult_test.csv")
    ev = m.evaluate(te, device="cpu")
    assert ev.accuracy > 0.86
    assert ev.auc > 0.92
    # save/load round-trip preserves oblique predictions exactly
    p1 = m.predict(te, device="cpu")
    m.save(str(tmp_path / "obl"))
    m2 = ydf.load_model(str(tmp_path / "obl"))
    np.testing.assert_array_equal(p1, m2.predict(te, device="cpu"))
    # serialize/deserialize too
    m3 = ydf.deserialize_model(ydf.serialize_model(m))
    np.testing.assert_array_equal(p1, m3.predict(te, device="cpu"))


def test_oblique_embed_cpp(tmp_path):
    """C++ codegen of an oblique model compiles and matches predictions."""
    import subprocess
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/adult_binary_class_gbdt_oblique")
    te = pd.read_csv(f"{BASE}/dataset/adult_test.csv").head(200)
    src = ydf.to_cpp(m, "obl")
    X = m._encode_features(te)
    F, N = X.shape
    main = (
        "#include <cstdio>\n" + src +
        "int main() { static float x[%d];\n" % F +
        "  FILE* f = fopen(\"x.bin\", \"rb\");\n"
        "  for (int i = 0; i < %d; ++i) {\n" % N +
        "    if (fread(x, sizeof(float), %d, f) != %d) return 1;\n"
        % (F, F) +
        "    printf(\"%.9g\\n\", obl_predict(x));\n"
        "  }\n  return 0;\n}\n")
    cpp = tmp_path / "m.cpp"
    cpp.write_text(main)
    exe = tmp_path / "m"
    subprocess.run(["g++", "-O1", "-o", str(exe), str(cpp)], check=True)
    (tmp_path / "x.bin").write_bytes(
        np.ascontiguousarray(X.T, dtype=np.float32).tobytes())
    out = subprocess.run([str(exe)], cwd=tmp_path, capture_output=True,
                         text=True, check=True)
    got = np.array([float(v) for v in out.stdout.split()])
    want = m.predict(te, device="cpu")
    np.testing.assert_allclose(got, want, atol=2e-5)


def test_export_roundtrip_rf(tmp_path, binary_data):
    """RF export to the reference format (random_forest_header.pb +
    classifier-distribution leaves), bit-exact round-trip, both
    winner-take-all settings."""
    for wta in (True, False):
        m = ydf.RandomForestLearner(
            label="label", num_trees=8, max_depth=8, winner_take_all=wta,
            compute_oob_performances=False).train(binary_data)
        p = str(tmp_path / f"rf{wta}")
        ydf.export_ydf_model(m, p)
        m2 = ydf.load_ydf_model(p)
        np.testing.assert_array_equal(m.predict(binary_data, device="cpu"),
                                      m2.predict(binary_data, device="cpu"))


def test_export_roundtrip_rf_regression(tmp_path):
    rng = np.random.RandomState(1)
    x = rng.randn(2000).astype(np.float32)
    d = {"x": x, "y": rng.randn(2000).astype(np.float32),
         "label": (2 * x).astype(np.float32)}
    m = ydf.RandomForestLearner(label="label", task=ydf.Task.REGRESSION,
                                num_trees=8,
                                compute_oob_performances=False).train(d)
    p = str(tmp_path / "rfreg")
    ydf.export_ydf_model(m, p)
    np.testing.assert_array_equal(m.predict(d, device="cpu"),
                                  ydf.load_ydf_model(p).predict(
                                      d, device="cpu"))


def test_export_roundtrip_oblique(tmp_path, binary_data):
    """Oblique conditions survive the reference wire format."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, max_depth=3, validation_ratio=0,
        split_axis="SPARSE_OBLIQUE").train(binary_data)
    assert (m.forest.cat_idx <= -2).sum() > 0
    p = str(tmp_path / "obl")
    ydf.export_ydf_model(m, p)
    m2 = ydf.load_ydf_model(p)
    assert (m2.forest.cat_idx <= -2).sum() > 0
    np.testing.assert_array_equal(m.predict(binary_data, device="cpu"),
                                  m2.predict(binary_data, device="cpu"))


def test_na_value_routing_golden_ranking():
    """The reference's na_value condition bits route MISSING inputs; the
    ranking model's golden predictions only reproduce when NaNs follow
    them (global imputation was off by up to 1.7 before)."""
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/synthetic_ranking_gbdt")
    assert m.forest.has_na_routing
    te = pd.read_csv(f"{BASE}/dataset/synthetic_ranking_test.csv")
    p = m.predict(te, device="cpu")
    g = pd.read_csv(
        f"{BASE}/prediction/synthetic_ranking_gbdt_test.csv")["LABEL"]
    assert np.abs(p - g.values).max() < 1e-4


def test_golden_isolation_forest():
    """Imported IF model vs the reference's sklearn-scored golden file
    (c(n) conventions differ slightly between implementations)."""
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/gaussians_anomaly_if")
    te = pd.read_csv(f"{BASE}/dataset/gaussians_test.csv")
    p = m.predict(te, device="cpu")
    g = np.loadtxt(f"{BASE}/prediction/gaussians_anomaly_if_skl.csv")
    assert np.abs(p - g).max() < 0.01
    assert np.corrcoef(p, g)[0, 1] > 0.999


def test_import_uplift_rf():
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/sim_pte_categorical_uplift_rf")
    assert m.task() == ydf.Task.CATEGORICAL_UPLIFT
    assert m.metadata.get("uplift_treatment") == "treat"
    te = pd.read_csv(f"{BASE}/dataset/sim_pte_test.csv")
    p = m.predict(te, device="cpu")
    assert -1.0 <= p.min() and p.max() <= 1.0
    assert p.std() > 0.01


def test_na_routing_trained_model_unaffected(binary_data, tmp_path):
    """Our own trained models keep imputation-era behavior (all na bits
    zero) and still round-trip."""
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=5,
                                        validation_ratio=0).train(
                                            binary_data)
    assert not m.forest.has_na_routing
    p1 = m.predict(binary_data, device="cpu")
    m.save(str(tmp_path / "m"))
    np.testing.assert_array_equal(
        p1, ydf.load_model(str(tmp_path / "m")).predict(binary_data,
                                                        device="cpu"))


def test_import_categorical_set_model_sst():
    """sst_binary_class_gbdt: bag-of-words CATEGORICAL_SET conditions
    (vocab 2001 > 256-bit masks); host set-walk serving reproduces the
    model's published quality and persists."""
    pd = pytest.importorskip("pandas")
    m = ydf.load_ydf_model(f"{BASE}/model/sst_binary_class_gbdt")
    assert m.forest.has_set_conditions
    te = pd.read_csv(f"{BASE}/dataset/sst_binary_test.csv")
    ev = m.evaluate(te)
    assert ev.accuracy > 0.78
    assert ev.auc > 0.86


def test_import_model_variants():
    """Every importable reference adult GBT variant loads and keeps
    published-range quality (32-category, integerized, numerical-only,
    v2 format, tuned)."""
    pd = pytest.importorskip("pandas")
    te = pd.read_csv(f"{BASE}/dataset/adult_test.csv")
    floors = {"adult_binary_class_gbdt_32cat": 0.86,
              "adult_binary_class_gbdt_integerized": 0.82,
              "adult_binary_class_gbdt_only_num": 0.82,
              "adult_binary_class_gbdt_v2": 0.86,
              "adult_binary_class_gbdt_tuned": 0.86}
    for name, floor in floors.items():
        m = ydf.load_ydf_model(f"{BASE}/model/{name}")
        assert m.evaluate(te).accuracy > floor, name
    for name in ("iris_multi_class_gbdt_v2", "abalone_regression_gbdt_v2",
                 "8bits_numerical_binary_class_gbdt"):
        assert ydf.load_ydf_model(f"{BASE}/model/{name}").num_trees() > 0
