"""Multitasker, feature selection, embed codegen, calibration extras."""
import ctypes
import os
import subprocess
import tempfile

import numpy as np
import pytest

import ydf_amd as ydf


def test_multitasker(binary_data, regression_data):
    d = dict(binary_data)
    d["reg_label"] = (2 * d["x1"] + d["x2"]).astype(np.float32)
    learner = ydf.MultitaskerLearner(
        tasks=[ydf.MultitaskItem(label="label",
                                 task=ydf.Task.CLASSIFICATION),
               ydf.MultitaskItem(label="reg_label",
                                 task=ydf.Task.REGRESSION)],
        num_trees=15)
    mm = learner.train(d)
    preds = mm.predict(d)
    assert set(preds) == {"label", "reg_label"}
    evs = mm.evaluate(d)
    assert evs["label"].accuracy > 0.85
    assert evs["reg_label"].rmse < 1.0


def test_backward_feature_selection(binary_data):
    sel = ydf.BackwardSelectionFeatureSelector(objective_metric="accuracy")
    learner = ydf.GradientBoostedTreesLearner(label="label", num_trees=15)
    logs = sel.run(learner, binary_data, binary_data)
    # x3 is pure noise in the fixture; x1/x2 carry the signal
    assert "x1" in logs.selected_features
    assert len(logs.iterations) >= 2


def test_embed_cpp_codegen(binary_data, tmp_path):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=10,
                                        validation_ratio=0).train(
                                            binary_data)
    src = ydf.to_cpp(m, "gen")
    cpp = tmp_path / "m.cpp"
    cpp.write_text(src + '\nextern "C" float gen_predict_c(const float* f)'
                   '{return gen_predict(f);}\n')
    so = str(tmp_path / "m.so")
    subprocess.run(["g++", "-O2", "-shared", "-fPIC", str(cpp), "-o", so],
                   check=True)
    lib = ctypes.CDLL(so)
    lib.gen_predict_c.restype = ctypes.c_float
    X = m._encode_features(binary_data)
    ref = m.predict(binary_data, device="cpu")
    for i in range(0, X.shape[1], 509):
        row = np.ascontiguousarray(X[:, i])
        p = lib.gen_predict_c(
            row.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
        assert abs(p - ref[i]) < 1e-5


def test_model_comparison_and_cis():
    """compare_models (reference metric/comparison.h) + closed-form and
    bootstrap confidence intervals (metric.h:150-177)."""
    import ydf_amd as ydf
    from ydf_amd.metric.metric import bootstrap_confidence_intervals

    rng = np.random.RandomState(0)
    n = 4000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(2 * x1 - x2 + 0.5 * rng.randn(n) > 0, "a", "b")}
    weak = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=2, validation_ratio=0).train(d)
    strong = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=50, validation_ratio=0).train(d)
    cmp = ydf.compare_models(weak, strong, d)
    assert cmp.p_value < 0.01  # strong significantly better
    assert cmp.metrics_2["accuracy"] > cmp.metrics_1["accuracy"]
    ev = strong.evaluate(d)
    lo, hi = ev.accuracy_ci95
    assert lo < ev.accuracy < hi
    lo, hi = ev.auc_ci95
    assert lo < ev.auc <= hi
    y = (np.asarray(d["label"]) == strong.label_classes[1]).astype(
        np.float32)
    cis = bootstrap_confidence_intervals(y, strong.predict(d),
                                         ydf.Task.CLASSIFICATION,
                                         n_samples=30)
    assert cis["accuracy"][0] <= ev.accuracy <= cis["accuracy"][1]
    # regression comparison path
    yr = (2 * x1 - x2).astype(np.float32)
    dr = {"x1": x1, "x2": x2, "label": yr}
    w = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=2,
        validation_ratio=0).train(dr)
    s = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=50,
        validation_ratio=0).train(dr)
    cmp = ydf.compare_models(w, s, dr)
    assert cmp.p_value < 0.01


def test_to_java_codegen():
    """Java embed codegen (reference serving/embed/java): structural
    checks — javac is not available in this image, so the C++ twin
    (compiled + compared elsewhere) anchors the shared emitter logic."""
    import ydf_amd as ydf

    d = ydf.generate_synthetic_dataset(num_examples=2000, num_numerical=4,
                                       num_categorical=1, seed=3)
    m = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=5, validation_ratio=0).train(d)
    src = ydf.to_java(m, "M")
    assert "public final class M" in src
    assert "public static float predict(float[] f)" in src
    assert src.count("private static float tree") == 5
    assert "Math.exp" in src  # sigmoid link
    assert src.count("{") == src.count("}")
    # multi-class emits predictMulti
    d3 = ydf.generate_synthetic_dataset(num_examples=1500, num_classes=3,
                                        seed=4)
    m3 = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=6, validation_ratio=0).train(d3)
    src3 = ydf.to_java(m3)
    assert "predictMulti" in src3
    assert src3.count("{") == src3.count("}")


def test_to_docker(tmp_path):
    """to_docker (PYDF export_docker analogue): generated FastAPI app
    serves the saved model (exercised in-process via TestClient)."""
    import sys

    import ydf_amd as ydf

    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    d = ydf.generate_synthetic_dataset(num_examples=1500, seed=10)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=5,
                                        validation_ratio=0).train(d)
    out = tmp_path / "serve"
    ydf.to_docker(m, str(out))
    for fn in ("Dockerfile", "main.py", "requirements.txt",
               "model/header.pb"):
        assert (out / fn).exists()
    sys.path.insert(0, str(out))
    try:
        import importlib

        main = importlib.import_module("main")
        importlib.reload(main)
        client = TestClient(main.app)
        assert client.get("/").json()["model"]
        example = {k: v[0].item() if hasattr(v[0], "item") else str(v[0])
                   for k, v in d.items() if k != "LABEL"}
        r = client.post("/predict", json=example).json()
        want = float(m.predict({k: v[:1] for k, v in d.items()},
                               device="cpu")[0])
        np.testing.assert_allclose(r["predictions"][0], want, rtol=1e-5)
    finally:
        sys.path.remove(str(out))
        sys.modules.pop("main", None)


def test_registry_usage_hooks_folds():
    """Small reference-parity utils: name->factory registry
    (registration.h), usage telemetry hooks (usage.h), fold generator
    (fold_generator.h)."""
    import ydf_amd as ydf

    events = []
    ydf.usage.register_on_inference(lambda **k: events.append(k))
    try:
        d = ydf.generate_synthetic_dataset(num_examples=400, seed=1)
        lrn = ydf.get_learner("GRADIENT_BOOSTED_TREES")(
            label="LABEL", num_trees=3, validation_ratio=0)
        m = lrn.train(d)
        m.predict(d)
        assert events and events[0]["num_examples"] == 400
    finally:
        ydf.usage.clear()
    with pytest.raises(KeyError):
        ydf.get_learner("NOPE")
    folds = ydf.generate_folds(103, 5, seed=2)
    assert sum(len(f) for f in folds) == 103
    assert len(np.unique(np.concatenate(folds))) == 103
    groups = np.repeat(np.arange(20), 5)
    gf = ydf.generate_folds(100, 4, groups=groups)
    for f in gf:  # whole groups stay together
        assert set(groups[f]) & set(
            groups[np.setdiff1d(np.arange(100), f)]) == set()


def test_weighted_evaluation():
    import ydf_amd as ydf

    rng = np.random.RandomState(0)
    n = 3000
    x = rng.randn(n).astype(np.float32)
    d = {"x": x, "label": np.where(x > 0, "a", "b"),
         "w": np.where(x > 0, 5.0, 1.0).astype(np.float32)}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=5, features=["x"],
        validation_ratio=0).train(d)
    ev_u = m.evaluate(d)
    ev_w = m.evaluate(d, weights="w")
    assert ev_u.accuracy != ev_w.accuracy or ev_u.loss != ev_w.loss
    # direct array form + regression path
    yr = (2 * x).astype(np.float32)
    dr = {"x": x, "label": yr}
    mr = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=5,
        validation_ratio=0).train(dr)
    w = np.abs(x) + 0.1
    ev = mr.evaluate(dr, weights=w)
    assert ev.rmse is not None


def test_engine_api_cpu():
    import ydf_amd as ydf

    d = ydf.generate_synthetic_dataset(num_examples=1500,
                                       num_numerical=5,
                                       num_categorical=0, num_boolean=0,
                                       seed=2)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=8,
                                        validation_ratio=0).train(d)
    assert set(m.list_compatible_engines()) == {"flat", "8bit", "binned8", "qs"}
    m.force_engine("qs")   # CPU predict falls back to the flat twin
    assert m.predict(d, device="cpu").shape == (1500,)
    with pytest.raises(ValueError):
        m.force_engine("bogus")
    d2 = ydf.generate_synthetic_dataset(num_examples=800,
                                        num_categorical=2, seed=3)
    m2 = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=4,
                                         validation_ratio=0).train(d2)
    assert m2.list_compatible_engines() == ["flat"]


def test_embed_cpp_routing_codegen(binary_data, tmp_path):
    """Table-driven (ROUTING) codegen — the default algorithm
    (reference embed.proto:38) — on a 1000-tree forest: compiles in
    O(nodes) data and matches model.predict."""
    m = ydf.RandomForestLearner(label="label", num_trees=1000,
                                max_depth=6,
                                winner_take_all=False).train(binary_data)
    src = ydf.to_cpp(m, "big")  # ROUTING is the default
    assert "big_roots" in src and "while ((fi" in src
    cpp = tmp_path / "big.cpp"
    cpp.write_text(src + '\nextern "C" float big_predict_c(const float* f)'
                   '{return big_predict(f);}\n')
    so = str(tmp_path / "big.so")
    subprocess.run(["g++", "-O1", "-shared", "-fPIC", str(cpp), "-o", so],
                   check=True, timeout=600)
    lib = ctypes.CDLL(so)
    lib.big_predict_c.restype = ctypes.c_float
    X = m._encode_features(binary_data)
    ref = m.predict(binary_data, device="cpu")
    for i in range(0, X.shape[1], 709):
        row = np.ascontiguousarray(X[:, i])
        p = lib.big_predict_c(
            row.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
        assert abs(p - ref[i]) < 1e-5, (i, p, ref[i])


def test_embed_routing_categorical_and_na(tmp_path):
    """ROUTING handles categorical masks + NA routing."""
    rng = np.random.RandomState(4)
    n = 6000
    cat = rng.choice(["a", "b", "c", "d", "e"], n)
    x = rng.randn(n).astype(np.float32)
    x[rng.rand(n) < 0.2] = np.nan
    y = np.where((np.isin(cat, ["a", "c"])) ^ (np.nan_to_num(x) > 0.5),
                 "p", "q")
    data = {"cat": cat, "x": x, "label": y}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=30, validation_ratio=0,
        missing_value_policy="LOCAL_IMPUTATION").train(data)
    src = ydf.to_cpp(m, "cn")
    cpp = tmp_path / "cn.cpp"
    cpp.write_text(src + '\nextern "C" float cn_predict_c(const float* f)'
                   '{return cn_predict(f);}\n')
    so = str(tmp_path / "cn.so")
    subprocess.run(["g++", "-O2", "-shared", "-fPIC", str(cpp), "-o", so],
                   check=True, timeout=300)
    lib = ctypes.CDLL(so)
    lib.cn_predict_c.restype = ctypes.c_float
    X = m._encode_features(data)
    ref = m.predict(data, device="cpu")
    for i in range(0, X.shape[1], 499):
        row = np.ascontiguousarray(X[:, i])
        p = lib.cn_predict_c(
            row.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
        assert abs(p - ref[i]) < 1e-5, (i, p, ref[i])


def test_to_js_codegen(binary_data, tmp_path):
    """JavaScript codegen (capability analogue of the reference JS port,
    port/javascript WASM inference): executed under node with
    prediction parity against model.predict."""
    import shutil

    if shutil.which("node") is None:
        pytest.skip("node not available")
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20,
                                        validation_ratio=0).train(
                                            binary_data)
    src = ydf.to_js(m, "gen")
    js = tmp_path / "model.js"
    X = m._encode_features(binary_data)
    idx = list(range(0, X.shape[1], 401))
    rows = [[float(v) for v in X[:, i]] for i in idx]
    driver = (src + "\nconst m = module.exports;\n"
              + f"const rows = {rows!r};\n"
              + "for (const r of rows) console.log(m.predict(r));\n")
    js.write_text(driver)
    r = subprocess.run(["node", str(js)], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    got = np.array([float(s) for s in r.stdout.split()])
    want = m.predict(binary_data, device="cpu")[idx]
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def test_to_js_multiclass_and_categorical(tmp_path):
    import shutil

    if shutil.which("node") is None:
        pytest.skip("node not available")
    rng = np.random.RandomState(5)
    n = 4000
    cat = rng.choice(["a", "b", "c", "d", "e"], n)
    x = rng.randn(n).astype(np.float32)
    y = np.where(np.isin(cat, ["a", "c"]), "u",
                 np.where(x > 0, "v", "w"))
    d = {"cat": cat, "x": x, "label": y}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=12,
                                        validation_ratio=0).train(d)
    src = ydf.to_js(m, "mc")
    X = m._encode_features(d)
    idx = list(range(0, n, 307))
    rows = [[float(v) for v in X[:, i]] for i in idx]
    driver = (src + "\nconst m = module.exports;\n"
              + f"const rows = {rows!r};\n"
              + "for (const r of rows) "
              "console.log(m.predict(r).join(','));\n")
    js = tmp_path / "mc.js"
    js.write_text(driver)
    r = subprocess.run(["node", str(js)], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    got = np.array([[float(v) for v in line.split(",")]
                    for line in r.stdout.strip().splitlines()])
    want = m.predict(d, device="cpu")[idx]
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_help_and_version():
    import ydf_amd as ydf

    assert "GRADIENT_BOOSTED_TREES" in ydf.help.learners()
    assert "num_trees" in ydf.help.hyperparameters(
        ydf.GradientBoostedTreesLearner)
    assert "csv:" in ydf.help.loading_data()
    assert ydf.version == ydf.__version__
    assert ydf.experimental.MultiLayerPerceptronLearner is not None


def test_learner_feature_selector_integration(binary_data):
    """learner(feature_selector=...) trains the final model on the
    selected subset and attaches the logs (PYDF integration)."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15,
        feature_selector=ydf.BackwardSelectionFeatureSelector(
            objective_metric="accuracy")).train(binary_data)
    logs = m.feature_selection_logs()
    assert logs is not None
    assert "x1" in logs.selected_features
    assert set(m.input_feature_names()) == set(logs.selected_features)
    assert m.evaluate(binary_data).accuracy > 0.9


def test_evaluation_binary_statistics(binary_data):
    """precision/recall/F1/FPR + per-threshold ROC characteristics
    (PYDF evaluation surface; reference metric Roc curves)."""
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=30,
                                        validation_ratio=0).train(
        binary_data)
    ev = m.evaluate(binary_data)
    assert 0.9 < ev.precision <= 1.0
    assert 0.9 < ev.recall <= 1.0
    assert 0.9 < ev.f1 <= 1.0
    assert 0.0 <= ev.false_positive_rate < 0.1
    ch = ev.characteristics[0]
    assert len(ch["fpr"]) == len(ch["tpr"]) == len(ch["thresholds"])
    assert (np.diff(ch["fpr"]) >= 0).all()  # monotone sweep
    assert (np.diff(ch["tpr"]) >= 0).all()
    # trapezoid integral of the curve must approximate the rank AUC
    auc_trap = float(np.trapz(ch["tpr"], ch["fpr"]))
    assert abs(auc_trap - ev.auc) < 0.01


def test_learner_name_and_input_feature_names(binary_data):
    """PYDF learner.learner_name + extract_input_feature_names."""
    ln = ydf.GradientBoostedTreesLearner(label="label")
    assert ln.learner_name == "GRADIENT_BOOSTED_TREES"
    assert ydf.RandomForestLearner(label="label").learner_name == \
        "RANDOM_FOREST"
    names = ln.extract_input_feature_names(binary_data)
    assert sorted(names) == ["x1", "x2", "x3"]
    ln2 = ydf.GradientBoostedTreesLearner(label="label",
                                          features=["x2", "nope"])
    assert ln2.extract_input_feature_names(binary_data) == ["x2"]


def test_log_book(tmp_path):
    """ydf.util.LogBook (reference util/log_book.py): SQLite-backed
    experiment tracking with default keys and superset filtering."""
    lb = ydf.util.LogBook(str(tmp_path / "lb"),
                          print_num_experiments=False,
                          default_keys={"project": "demo"})
    key = {"param1": 1, "param2": "abc"}
    assert not lb.exist(key)
    lb.add(key, {"accuracy": 0.9, "obs": [1, 2, 3]})
    assert lb.exist(key)
    assert lb.count_key(key) == 1
    assert lb.num_experiments() == 1
    with pytest.raises(ValueError):
        lb.add(key, {"accuracy": 0.91})  # duplicate key
    with pytest.raises(ValueError):
        lb.add({"id": 1}, {})  # reserved key
    lb.add({"param1": 2, "param2": "x"}, {"accuracy": 0.8})
    df = lb.to_dataframe()
    assert len(df) == 2 and "accuracy" in df.columns
    assert (df["project"] == "demo").all()
    assert len(lb.to_dataframe({"param1": 1})) == 1
    # reopen from disk
    lb2 = ydf.util.LogBook(str(tmp_path / "lb"),
                           print_num_experiments=False,
                           default_keys={"project": "demo"})
    assert lb2.num_experiments() == 2


def test_util_tf_record_roundtrip(tmp_path):
    """ydf.util.read_tf_record / write_tf_record."""
    cols = {"x": np.arange(5, dtype=np.float32),
            "name": np.array(["a", "b", "c", "d", "e"])}
    p = str(tmp_path / "data.tfrecord")
    ydf.util.write_tf_record(cols, p)
    back = ydf.util.read_tf_record(p)
    np.testing.assert_allclose(back["x"], cols["x"])
    assert list(back["name"].astype(str)) == list(cols["name"])


def test_evaluation_reference_surface(binary_data, regression_data):
    """Reference Evaluation surface: confusion_matrix with class names,
    num_examples_weighted, Characteristic precision/recall accessors +
    precision_at_recall, html(), regression bootstrap RMSE CI."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=30, validation_ratio=0.0).train(
        binary_data)
    ev = m.evaluate(binary_data)
    cm = ev.confusion_matrix
    assert cm.matrix.shape == (2, 2)
    assert set(cm.classes) == set(m.label_classes)
    assert cm.value(cm.classes[0], cm.classes[0]) == cm.matrix[0, 0]
    assert ev.num_examples_weighted == ev.num_examples
    ch = ev.characteristics[0]
    assert len(ch.precisions) == len(ch.recalls) == len(ch.thresholds)
    assert 0.0 <= ch.precision_at_recall(0.5) <= 1.0
    assert ch.precision_at_recall(0.0) == 1.0
    assert ch.roc_auc == ev.auc
    assert "<table>" in ev.html()

    mr = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=30,
        validation_ratio=0.0).train(regression_data)
    evr = mr.evaluate(regression_data)
    lo, hi = evr.rmse_ci95_bootstrap
    assert lo <= evr.rmse <= hi


def test_ranking_map_mrr():
    rng = np.random.RandomState(3)
    n = 4000
    g = np.repeat(np.arange(n // 8), 8)
    x = rng.randn(n).astype(np.float32)
    rel = np.clip((x + rng.randn(n) * 0.3) * 2, 0, 4).astype(np.float32)
    d = {"x": x, "rel": rel, "g": g}
    m = ydf.GradientBoostedTreesLearner(
        label="rel", task=ydf.Task.RANKING, ranking_group="g",
        num_trees=30, validation_ratio=0.0).train(d)
    ev = m.evaluate(d)
    assert ev.ndcg > 0.75
    assert 0.0 < ev.map <= 1.0
    assert 0.0 < ev.mrr <= 1.0


def test_tuner_optimize_metric(binary_data):
    """RandomSearchTuner(optimize_metric=...) (PYDF OptimizeMetric):
    trial selection by evaluated metric instead of validation loss."""
    from ydf_amd.learner.tuner import OptimizeMetric

    t = ydf.RandomSearchTuner(num_trials=3, seed=7,
                              optimize_metric=OptimizeMetric.ACCURACY)
    t.choice("max_depth", [2, 4])
    t.choice("shrinkage", [0.05, 0.15])
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, tuner=t).train(binary_data)
    assert m.tuner_logs is not None
    assert len(m.tuner_logs.trials) == 3
    best = m.tuner_logs.best_trial
    # accuracy objective: scores are accuracies in [0, 1]
    assert 0.5 < best.score <= 1.0


def test_evaluate_keyword_surface(binary_data, regression_data):
    """Reference evaluate() keyword surface: weighted=False, label/task
    override, bootstrapping CIs, use_slow_engine, truncations."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, validation_ratio=0.0).train(
        binary_data)
    ev = m.evaluate(binary_data, weighted=False, use_slow_engine=True,
                    num_threads=2)
    assert ev.accuracy > 0.9
    evb = m.evaluate(binary_data, bootstrapping=50)
    lo, hi = evb.bootstrap_cis["accuracy"]
    assert lo <= evb.accuracy <= hi

    # label override: same values under a different column name
    data2 = dict(binary_data)
    data2["y2"] = binary_data["label"]
    del data2["label"]
    ev2 = m.evaluate(data2, label="y2")
    assert abs(ev2.accuracy - ev.accuracy) < 1e-9

    # task override: regression metrics on a regression model under
    # an explicitly passed task
    mr = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=20,
        validation_ratio=0.0).train(regression_data)
    evr = mr.evaluate(regression_data, task=ydf.Task.REGRESSION)
    assert evr.rmse is not None


def test_ranking_truncations():
    rng = np.random.RandomState(5)
    n = 2000
    g = np.repeat(np.arange(n // 10), 10)
    x = rng.randn(n).astype(np.float32)
    rel = np.clip((x + rng.randn(n) * 0.4) * 2, 0, 4).astype(np.float32)
    d = {"x": x, "rel": rel, "g": g}
    m = ydf.GradientBoostedTreesLearner(
        label="rel", task=ydf.Task.RANKING, ranking_group="g",
        num_trees=20, validation_ratio=0.0).train(d)
    e5 = m.evaluate(d)
    e10 = m.evaluate(d, ndcg_truncation=10, mrr_truncation=10,
                     map_truncation=10)
    assert e5.ndcg != e10.ndcg  # truncation changes the metric
    assert 0 < e10.mrr <= 1 and 0 < e10.map <= 1
    # group override by explicit column name
    e_g = m.evaluate(d, group="g")
    assert abs(e_g.ndcg - e5.ndcg) < 1e-12


def test_analyze_keyword_surface(binary_data):
    """Reference analyze() kwargs: sampling, plot toggles, SHAP
    summary importance, permutation rounds."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, validation_ratio=0.0).train(
        binary_data)
    a = m.analyze(binary_data, sampling=0.3, num_bins=10,
                  permutation_variable_importance_rounds=2,
                  maximum_duration=30, num_threads=2)
    assert "MEAN_ABS_SHAP" in a.variable_importances
    assert a.partial_dependences and a.conditional_expectations
    a2 = m.analyze(binary_data, partial_dependence_plot=False,
                   conditional_expectation_plot=False,
                   shap_values=False,
                   permutation_variable_importance=False)
    assert not a2.partial_dependences
    assert not a2.conditional_expectations
    assert "MEAN_ABS_SHAP" not in a2.variable_importances
    assert "MEAN_DECREASE_IN_ACCURACY" not in a2.variable_importances


def test_serving_session_requires_gpu(binary_data):
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=5, validation_ratio=0.0).train(
        binary_data)
    if not __import__("torch").cuda.is_available():
        with pytest.raises(ValueError, match="GPU"):
            m.serving_session(64, device="cpu")
