"""End-to-end learner quality tests (reference analogue: the
TrainAndTestTester metric-margin checks, utils/test_utils.h:428)."""
import numpy as np
import pytest

import ydf_amd as ydf


def test_gbt_binary_quality(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=100).train(
        binary_data)
    ev = m.evaluate(binary_data)
    assert ev.accuracy > 0.93
    assert ev.auc > 0.97
    assert ev.loss < 0.2


def test_gbt_regression_quality(regression_data):
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=150).train(
            regression_data)
    ev = m.evaluate(regression_data)
    assert ev.rmse < 0.35


def test_gbt_multiclass(binary_data):
    rng = np.random.RandomState(2)
    n = 5000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0.5, "a", np.where(x2 > 0, "b", "c"))
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=50).train(d)
    ev = m.evaluate(d)
    assert ev.accuracy > 0.98
    p = m.predict(d)
    assert p.shape == (n, 3)
    np.testing.assert_allclose(p.sum(axis=1), 1.0, atol=1e-4)


def test_gbt_early_stopping_truncates(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=300,
                                        validation_ratio=0.15).train(
                                            binary_data)
    assert m.num_trees() <= 300
    assert m.training_logs, "validation logs missing"


def test_gbt_no_validation(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20,
                                        validation_ratio=0.0).train(
                                            binary_data)
    assert m.num_trees() == 20


def test_gbt_subsample(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=60,
                                        subsample=0.5).train(binary_data)
    ev = m.evaluate(binary_data)
    assert ev.accuracy > 0.9


def test_rf_binary(binary_data):
    m = ydf.RandomForestLearner(label="label", num_trees=50,
                                max_depth=12).train(binary_data)
    ev = m.evaluate(binary_data)
    assert ev.accuracy > 0.93
    p = m.predict(binary_data)
    assert p.min() >= 0.0 and p.max() <= 1.0


def test_rf_regression(regression_data):
    m = ydf.RandomForestLearner(label="label", task=ydf.Task.REGRESSION,
                                num_trees=50).train(regression_data)
    ev = m.evaluate(regression_data)
    assert ev.rmse < 1.0


def test_rf_multiclass():
    rng = np.random.RandomState(4)
    n = 4000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0.5, "a", np.where(x2 > 0, "b", "c"))
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.RandomForestLearner(label="label", num_trees=30).train(d)
    ev = m.evaluate(d)
    assert ev.accuracy > 0.95


def test_cart(binary_data):
    m = ydf.CartLearner(label="label").train(binary_data)
    assert m.num_trees() == 1
    ev = m.evaluate(binary_data)
    assert ev.accuracy > 0.9


def test_isolation_forest():
    rng = np.random.RandomState(7)
    inliers = rng.randn(2000, 2)
    outliers = rng.randn(50, 2) * 0.3 + 6.0
    X = np.concatenate([inliers, outliers])
    d = {"a": X[:, 0].astype(np.float32), "b": X[:, 1].astype(np.float32)}
    m = ydf.IsolationForestLearner(num_trees=100).train(d)
    s = m.predict(d)
    assert s.shape == (2050,)
    # outliers must score clearly higher
    assert s[2000:].mean() > s[:2000].mean() + 0.1
    # AUC of anomaly detection
    from ydf_amd.metric.metric import roc_auc

    labels = np.zeros(2050, dtype=bool)
    labels[2000:] = True
    assert roc_auc(labels, s) > 0.95


def test_deterministic_same_seed(binary_data):
    m1 = ydf.GradientBoostedTreesLearner(label="label", num_trees=10,
                                         validation_ratio=0).train(
                                             binary_data)
    m2 = ydf.GradientBoostedTreesLearner(label="label", num_trees=10,
                                         validation_ratio=0).train(
                                             binary_data)
    np.testing.assert_array_equal(m1.forest.feat, m2.forest.feat)
    np.testing.assert_array_equal(m1.forest.thr, m2.forest.thr)


def test_adult_gbt_quality(adult_paths):
    pd = pytest.importorskip("pandas")
    tr, te = adult_paths
    m = ydf.GradientBoostedTreesLearner(label="income").train(
        pd.read_csv(tr))
    ev = m.evaluate(pd.read_csv(te))
    # reference GBT reaches ~0.873 accuracy / ~0.929 AUC on adult
    assert ev.accuracy > 0.865
    assert ev.auc > 0.92


def test_feature_subset(binary_data):
    m = ydf.GradientBoostedTreesLearner(
        label="label", features=["x1", "x2"], num_trees=20).train(
            binary_data)
    assert m.input_feature_names() == ["x1", "x2"]
    m.predict({"x1": binary_data["x1"], "x2": binary_data["x2"]})


def test_categorical_set_splits():
    """Categorical features must use set-splits (mask conditions), and beat
    what ordinal encoding could do on a label-ordered category problem."""
    rng = np.random.RandomState(8)
    n = 20000
    # category c's target probability is NOT monotone in the code, so a
    # single ordinal threshold cannot separate well, a set can
    cats = rng.randint(0, 12, n)
    probs = np.array([0.9, 0.1, 0.8, 0.2, 0.95, 0.05,
                      0.85, 0.15, 0.9, 0.1, 0.8, 0.2])
    y = rng.rand(n) < probs[cats]
    d = {"c": np.array([f"cat{v}" for v in cats]),
         "x": rng.randn(n).astype(np.float32),
         "label": np.where(y, "p", "n")}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=30,
                                        max_depth=3).train(d)
    assert (m.forest.cat_idx >= 0).any(), "no categorical set-splits used"
    ev = m.evaluate(d)
    assert ev.accuracy > 0.82  # bayes ~0.85; ordinal threshold ~0.6


def test_categorical_model_roundtrip(tmp_path):
    rng = np.random.RandomState(9)
    n = 3000
    cats = rng.randint(0, 6, n)
    y = cats % 2 == 0
    d = {"c": np.array([f"v{v}" for v in cats]),
         "label": np.where(y, "a", "b")}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=10,
                                        validation_ratio=0).train(d)
    p = str(tmp_path / "catm")
    m.save(p)
    m2 = ydf.load_model(p)
    np.testing.assert_allclose(m.predict(d), m2.predict(d), rtol=1e-6)
    assert m.evaluate(d).accuracy > 0.99


def test_checkpoint_resume_exact(tmp_path, binary_data):
    """Resumed training must produce the same model as a straight run up to
    float-ulp re-summation of margins (reference try_resume_training,
    gradient_boosted_trees.cc:1403-1443)."""
    wd = str(tmp_path / "wd")
    kw = dict(label="label", validation_ratio=0.1, early_stopping="NONE")
    ydf.GradientBoostedTreesLearner(
        num_trees=15, working_dir=wd, **kw).train(binary_data)
    m2 = ydf.GradientBoostedTreesLearner(
        num_trees=30, working_dir=wd, resume_training=True,
        **kw).train(binary_data)
    assert m2.num_trees() == 30
    m3 = ydf.GradientBoostedTreesLearner(num_trees=30, **kw).train(
        binary_data)
    p2, p3 = m2.predict(binary_data), m3.predict(binary_data)
    # margins are re-summed on resume -> ulp-level differences may flip
    # near-tie splits for a handful of examples
    assert np.mean(np.abs(p2 - p3) < 1e-4) > 0.99
    assert np.abs(p2 - p3).max() < 0.2


def test_maximum_training_duration(binary_data):
    import time

    t0 = time.time()
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=100000, validation_ratio=0,
        maximum_training_duration_seconds=1.5).train(binary_data)
    assert time.time() - t0 < 15
    assert 1 <= m.num_trees() < 100000


def test_variable_importances_and_analyze(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20).train(
        binary_data)
    vi = m.variable_importances()
    assert "SUM_SCORE" in vi and "NUM_NODES" in vi
    # x1 carries most signal in the fixture
    assert vi["SUM_SCORE"][0][1] == "x1"
    an = m.analyze(binary_data)
    assert "MEAN_DECREASE_IN_ACCURACY" in an.variable_importances
    assert len(an.partial_dependences) == 3
    assert an._repr_html_()


def test_example_weights(binary_data):
    rng = np.random.RandomState(3)
    d = dict(binary_data)
    d["w"] = (rng.rand(len(d["x1"])) + 0.5).astype(np.float32)
    m = ydf.GradientBoostedTreesLearner(label="label", weights="w",
                                        num_trees=30).train(d)
    assert "w" not in m.input_feature_names()
    assert m.evaluate(binary_data).accuracy > 0.9


def test_random_search_tuner(binary_data):
    tuner = ydf.RandomSearchTuner(num_trials=3)
    tuner.choice("shrinkage", [0.05, 0.2])
    tuner.choice("max_depth", [3, 5])
    m = ydf.GradientBoostedTreesLearner(label="label", tuner=tuner,
                                        num_trees=20).train(binary_data)
    assert m.tuner_logs is not None
    assert len(m.tuner_logs.trials) == 3
    assert m.tuner_logs.best_trial.score >= max(
        t.score for t in m.tuner_logs.trials) - 1e-9


def test_parallel_tuner_matches_sequential(binary_data):
    """parallel_trials=4 (one trial per process slot; one per GPU when
    present — reference distributed HPO analogue) must sample the same
    trials as sequential search and pick an equally good winner, and
    the tuning logs must show up in describe()."""
    def make_tuner(par):
        t = ydf.RandomSearchTuner(num_trials=4, seed=7,
                                  parallel_trials=par)
        t.choice("shrinkage", [0.05, 0.2])
        t.choice("max_depth", [3, 5])
        return t

    kw = dict(label="label", num_trees=10, device="cpu")
    m_par = ydf.GradientBoostedTreesLearner(
        tuner=make_tuner(4), **kw).train(binary_data)
    m_seq = ydf.GradientBoostedTreesLearner(
        tuner=make_tuner(1), **kw).train(binary_data)
    assert len(m_par.tuner_logs.trials) == 4
    hp_par = [t.hyperparameters for t in m_par.tuner_logs.trials]
    hp_seq = [t.hyperparameters for t in m_seq.tuner_logs.trials]
    assert hp_par == hp_seq  # same seed -> same sampled trials
    np.testing.assert_allclose(m_par.tuner_logs.best_trial.score,
                               m_seq.tuner_logs.best_trial.score,
                               rtol=1e-5)
    assert "tuning: 4 trials" in m_par.describe()


def test_cross_validation(binary_data):
    ev = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15).cross_validation(binary_data, folds=3)
    assert ev.accuracy > 0.9
    assert ev.num_examples == len(binary_data["x1"])


def test_goss_sampling(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=40,
                                        sampling_method="GOSS").train(
                                            binary_data)
    assert m.evaluate(binary_data).accuracy > 0.9


def test_rf_oob_and_winner_take_all(binary_data):
    m = ydf.RandomForestLearner(label="label", num_trees=30,
                                max_depth=10).train(binary_data)
    ev = m.self_evaluation()
    assert ev is not None and ev.accuracy > 0.9
    # WTA leaves are votes in {0,1}; mean over trees stays in [0,1]
    p = m.predict(binary_data)
    assert p.min() >= 0 and p.max() <= 1


def test_custom_regression_loss(regression_data):
    def gh(labels, preds):
        r = preds - labels
        return np.clip(r, -1, 1), np.ones_like(r)

    L = ydf.RegressionLoss(
        gradient_and_hessian=gh,
        initial_predictions=lambda y, w: float(np.median(y)),
        loss=lambda y, p, w: float(np.abs(p - y).mean()))
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, loss=L, num_trees=60,
        shrinkage=0.3).train(regression_data)
    assert m.evaluate(regression_data).rmse < 1.0


def test_custom_binary_loss(binary_data):
    def gh(labels, preds):
        p = 1.0 / (1.0 + np.exp(-preds))
        return p - labels, np.maximum(p * (1 - p), 1e-6)

    L = ydf.BinaryClassificationLoss(gradient_and_hessian=gh)
    m = ydf.GradientBoostedTreesLearner(label="label", loss=L,
                                        num_trees=40).train(binary_data)
    assert m.evaluate(binary_data).accuracy > 0.9


def _ranking_data(seed=0, Q=500, M=12):
    rng = np.random.RandomState(seed)
    n = Q * M
    group = np.repeat(np.arange(Q), M)
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    true = 2 * x1 + x2
    rel = np.zeros(n, dtype=np.float32)
    for q in range(Q):
        order = np.argsort(np.argsort(-true[q * M:(q + 1) * M]))
        rel[q * M:(q + 1) * M] = np.clip(4 - order // 3, 0, 4)
    return {"x1": x1, "x2": x2, "g": group, "label": rel}


def test_ranking_lambdamart_ndcg():
    d = _ranking_data()
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.RANKING, ranking_group="g",
        num_trees=50).train(d)
    ev = m.evaluate(d)
    assert ev.ndcg is not None and ev.ndcg > 0.95
    assert "g" not in m.input_feature_names()


def test_ranking_model_roundtrip(tmp_path):
    d = _ranking_data(Q=100)
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.RANKING, ranking_group="g",
        num_trees=10, validation_ratio=0).train(d)
    p = str(tmp_path / "rankm")
    m.save(p)
    m2 = ydf.load_model(p)
    np.testing.assert_allclose(m.predict(d), m2.predict(d), rtol=1e-6)
    assert m2.evaluate(d).ndcg is not None


def test_tree_shap_efficiency(regression_data):
    """Sum of SHAP values + bias must equal the prediction (the Shapley
    efficiency property; reference utils/shap.h TreeSHAP)."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=25,
        validation_ratio=0).train(regression_data)
    shap = m.predict_shap(regression_data)
    phi = np.stack([shap[k] for k in m.input_feature_names()], axis=1)
    total = phi.sum(axis=1) + shap["__BIAS__"]
    np.testing.assert_allclose(total, m.predict(regression_data), atol=1e-4)


def test_tree_shap_binary_margin(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20).train(
        binary_data)
    shap = m.predict_shap(binary_data)
    phi = np.stack([shap[k] for k in m.input_feature_names()], axis=1)
    margin = phi.sum(axis=1) + shap["__BIAS__"]
    prob = 1.0 / (1.0 + np.exp(-margin))
    np.testing.assert_allclose(prob, m.predict(binary_data), atol=1e-4)
    # x1 dominates the fixture
    means = {k: np.abs(v).mean() for k, v in shap.items()
             if k != "__BIAS__"}
    assert max(means, key=means.get) == "x1"


def test_monotonic_constraint():
    """Predictions must be non-decreasing in a monotonic(+1) feature
    (reference monotonic constraints, decision_tree.proto)."""
    rng = np.random.RandomState(0)
    n = 6000
    x = rng.randn(n).astype(np.float32)
    z = rng.randn(n).astype(np.float32)
    y = (x + 0.3 * np.sin(5 * x) + z).astype(np.float32)
    d = {"x": x, "z": z, "label": y}
    feats = [ydf.Feature("x", monotonic=ydf.Monotonic.INCREASING),
             ydf.Feature("z")]
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, features=feats,
        num_trees=60, validation_ratio=0).train(d)
    grid = np.linspace(-3, 3, 300).astype(np.float32)
    p = m.predict({"x": grid, "z": np.zeros_like(grid)})
    assert np.diff(p).min() >= -1e-6
    assert m.evaluate(d).rmse < 0.5


def test_cart_pruning_reduces_overfit():
    rng = np.random.RandomState(0)
    n = 8000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0, "p", "n")
    flip = rng.rand(n) < 0.25
    y = np.where(flip, np.where(y == "p", "n", "p"), y)
    d = {"x1": x1, "x2": x2, "label": y}
    m_pruned = ydf.CartLearner(label="label").train(d)
    m_raw = ydf.CartLearner(label="label", validation_ratio=0).train(d)
    assert m_pruned.num_nodes() < m_raw.num_nodes() / 2
    x1t = rng.randn(4000).astype(np.float32)
    dt = {"x1": x1t, "x2": rng.randn(4000).astype(np.float32),
          "label": np.where(x1t > 0, "p", "n")}
    assert m_pruned.evaluate(dt).accuracy > m_raw.evaluate(dt).accuracy


def test_poisson_loss():
    rng = np.random.RandomState(0)
    n = 6000
    x = rng.randn(n).astype(np.float32)
    lam = np.exp(0.5 + x)
    y = rng.poisson(lam).astype(np.float32)
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, loss="POISSON",
        num_trees=60).train({"x": x, "label": y})
    p = m.predict({"x": x})
    assert np.corrcoef(p, lam)[0, 1] > 0.95
    assert p.min() >= 0  # exp link


def test_mae_loss_robust_to_outliers():
    rng = np.random.RandomState(0)
    n = 6000
    x = rng.randn(n).astype(np.float32)
    y = (2 * x).astype(np.float32)
    y[:150] += 100.0
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, loss="MEAN_AVERAGE_ERROR",
        num_trees=150, shrinkage=0.3, validation_ratio=0).train(
            {"x": x, "label": y})
    p = m.predict({"x": x})
    assert np.median(np.abs(p - 2 * x)) < 0.5


def _rotated_data(n=8000, seed=0):
    rng = np.random.RandomState(seed)
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    return {"x1": x1, "x2": x2,
            "label": np.where(x1 + x2 > 0, "p", "n")}


def test_oblique_gbt_beats_axis_aligned(tmp_path):
    """A diagonal decision boundary: oblique splits must (a) appear in the
    model and (b) beat axis-aligned trees of the same budget (reference
    SparseObliqueSplit behavior)."""
    tr = _rotated_data(8000, 0)
    te = _rotated_data(2000, 1)
    kw = dict(label="label", num_trees=20, max_depth=3, validation_ratio=0)
    acc_ax = ydf.GradientBoostedTreesLearner(**kw).train(tr).evaluate(
        te).accuracy
    m = ydf.GradientBoostedTreesLearner(
        split_axis="SPARSE_OBLIQUE", **kw).train(tr)
    acc_ob = m.evaluate(te).accuracy
    assert (m.forest.cat_idx <= -2).sum() > 0
    assert acc_ob > acc_ax
    assert acc_ob > 0.99
    # persistence round-trip keeps oblique predictions exactly
    p1 = m.predict(te, device="cpu")
    m.save(str(tmp_path / "obl"))
    m2 = ydf.load_model(str(tmp_path / "obl"))
    np.testing.assert_array_equal(p1, m2.predict(te, device="cpu"))


def test_oblique_gbt_validation_and_subsample():
    """Oblique + internal validation split (routes the valid rows through
    per-level replayed projections) + row subsampling (routes
    out-of-sample rows)."""
    tr = _rotated_data(6000, 2)
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, max_depth=3, validation_ratio=0.15,
        subsample=0.6, split_axis="SPARSE_OBLIQUE").train(tr)
    assert m.evaluate(_rotated_data(2000, 3)).accuracy > 0.98
    assert m.training_logs  # validation loss was tracked


def test_oblique_rf():
    tr = _rotated_data(6000, 4)
    m = ydf.RandomForestLearner(
        label="label", num_trees=10, max_depth=8,
        split_axis="SPARSE_OBLIQUE",
        compute_oob_performances=False).train(tr)
    assert (m.forest.cat_idx <= -2).sum() > 0
    assert m.evaluate(_rotated_data(2000, 5)).accuracy > 0.97


def test_oblique_with_categorical_features():
    """Categorical features keep set-splits; projections draw only from
    the numerical columns."""
    rng = np.random.RandomState(7)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    c = rng.randint(0, 6, n)
    y = (x1 + x2 > 0) ^ (c % 3 == 0)
    d = {"x1": x1, "x2": x2, "c": np.array([f"v{v}" for v in c]),
         "label": np.where(y, "p", "n")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=25, max_depth=4, validation_ratio=0,
        split_axis="SPARSE_OBLIQUE").train(d)
    assert m.evaluate(d).accuracy > 0.95
    f = m.forest
    # oblique conditions never reference the categorical column
    ci = f.dataspec_cat_col if hasattr(f, "dataspec_cat_col") else None
    cat_pos = [i for i, cspec in enumerate(m.dataspec.feature_columns)
               if cspec.name == "c"][0]
    assert not np.any(f.obl_attr == cat_pos)


def test_focal_loss_binary():
    """BINARY_FOCAL_LOSS (reference loss_imp_binary_focal.cc): trains a
    usable classifier on an imbalanced problem; sigmoid activation."""
    rng = np.random.RandomState(0)
    n = 8000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where((2 * x1 - x2 + 0.3 * rng.randn(n)) > 1.8, "pos", "neg")
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=40, loss="BINARY_FOCAL_LOSS",
        focal_loss_alpha=0.75, validation_ratio=0.1).train(d)
    ev = m.evaluate(d)
    assert ev.auc > 0.99
    p = m.predict(d)
    assert 0.0 <= p.min() and p.max() <= 1.0  # sigmoid applied
    with pytest.raises(ValueError):
        ydf.GradientBoostedTreesLearner(
            label="label", loss="BINARY_FOCAL_LOSS").train(
                {"x": np.arange(30, dtype=np.float32),
                 "label": np.array(["a", "b", "c"] * 10)})


def test_xe_ndcg_ranking():
    """XE_NDCG_MART (reference loss_imp_cross_entropy_ndcg.cc)."""
    rng = np.random.RandomState(1)
    q = np.repeat(np.arange(200), 10)
    rel = rng.randint(0, 5, 2000).astype(np.float32)
    d = {"q": q, "rel": rel,
         "f1": (rel + rng.randn(2000)).astype(np.float32),
         "f2": rng.randn(2000).astype(np.float32)}
    m = ydf.GradientBoostedTreesLearner(
        label="rel", ranking_group="q", task=ydf.Task.RANKING,
        num_trees=30, loss="XE_NDCG_MART").train(d)
    assert m.evaluate(d).ndcg > 0.9


def test_dart():
    """DART forest extraction: dropout + retroactive rescale. With one
    tree no dropout can occur, so DART == MART exactly; with many trees
    the model stays accurate."""
    rng = np.random.RandomState(2)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(2 * x1 - x2 + 0.5 * x1 * x2 > 0, "a", "b")}
    kw = dict(label="label", validation_ratio=0)
    m1 = ydf.GradientBoostedTreesLearner(
        num_trees=1, forest_extraction="DART", dart_dropout=0.5, **kw
    ).train(d)
    m2 = ydf.GradientBoostedTreesLearner(num_trees=1, **kw).train(d)
    np.testing.assert_allclose(m1.predict(d), m2.predict(d), atol=1e-6)
    m = ydf.GradientBoostedTreesLearner(
        num_trees=60, forest_extraction="DART", dart_dropout=0.1,
        validation_ratio=0.1, label="label").train(d)
    assert m.evaluate(d).accuracy > 0.98


def test_rf_honest_trees():
    """Honest trees (reference Honest message, decision_tree.proto): leaf
    values from held-out rows; quality stays competitive."""
    rng = np.random.RandomState(3)
    n = 4000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(2 * x1 - x2 + 0.3 * rng.randn(n) > 0, "a", "b")}
    m = ydf.RandomForestLearner(label="label", num_trees=25, max_depth=8,
                                honest=True).train(d)
    assert m.evaluate(d).accuracy > 0.93
    # fixed separation reuses one honest split across trees
    m2 = ydf.RandomForestLearner(label="label", num_trees=10, max_depth=6,
                                 honest=True,
                                 honest_fixed_separation=True).train(d)
    assert m2.evaluate(d).accuracy > 0.9


def test_rf_oob_permutation_importances():
    """OOB permutation variable importances (reference
    random_forest.cc:1411): informative features must outrank noise."""
    rng = np.random.RandomState(4)
    n = 3000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    noise = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2, "noise": noise,
         "label": np.where(2 * x1 - x2 + 0.3 * rng.randn(n) > 0, "a", "b")}
    m = ydf.RandomForestLearner(
        label="label", num_trees=20, max_depth=8,
        compute_oob_variable_importances=True).train(d)
    vi = m.variable_importances()["MEAN_DECREASE_IN_ACCURACY"]
    ranked = [name for _, name in vi]
    assert ranked[0] == "x1"
    assert ranked[-1] == "noise"
    scores = {name: s for s, name in vi}
    assert scores["x1"] > 0.1
    assert abs(scores["noise"]) < 0.05


def test_rf_bootstrap_ratio_and_max_duration():
    import time as _time

    d = ydf.generate_synthetic_dataset(num_examples=3000, seed=5)
    m = ydf.RandomForestLearner(label="LABEL", num_trees=5,
                                bootstrap_size_ratio=0.5,
                                compute_oob_performances=False).train(d)
    assert m.num_trees() == 5
    t0 = _time.monotonic()
    m2 = ydf.RandomForestLearner(
        label="LABEL", num_trees=100000,
        maximum_training_duration_seconds=1.0,
        compute_oob_performances=False).train(d)
    assert _time.monotonic() - t0 < 10
    assert 0 < m2.num_trees() < 100000


def test_analyze_prediction():
    d = ydf.generate_synthetic_dataset(num_examples=2000, seed=6)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=10,
                                        validation_ratio=0).train(d)
    one = {k: v[:1] for k, v in d.items()}
    ap = m.analyze_prediction(one)
    assert "__BIAS__" in ap
    assert set(m.input_feature_names()) <= set(ap)
    # contributions + bias reconstruct the margin (SHAP completeness)
    import torch
    X = torch.from_numpy(m._encode_features(one))
    margin = float(m.predict_margin(X)[0, 0])
    total = sum(float(v) for v in ap.values())
    np.testing.assert_allclose(total, margin, rtol=1e-3, atol=1e-3)


def test_selgb_sampling_ranking():
    rng = np.random.RandomState(1)
    q = np.repeat(np.arange(300), 10)
    rel = (rng.rand(3000) < 0.15) * rng.randint(1, 5, 3000)
    d = {"q": q, "rel": rel.astype(np.float32),
         "f1": (rel + rng.randn(3000)).astype(np.float32),
         "f2": rng.randn(3000).astype(np.float32)}
    m = ydf.GradientBoostedTreesLearner(
        label="rel", ranking_group="q", task=ydf.Task.RANKING,
        num_trees=30, sampling_method="SELGB",
        selective_gradient_boosting_ratio=0.2).train(d)
    assert m.evaluate(d).ndcg > 0.9


def test_hyperparameter_templates():
    d = ydf.generate_synthetic_dataset(num_examples=2500, seed=12)
    m = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=10, validation_ratio=0,
        hyperparameter_template="benchmark_rank1").train(d)
    assert (m.forest.cat_idx <= -2).sum() > 0  # template turns on oblique
    m2 = ydf.RandomForestLearner(
        label="LABEL", num_trees=5,
        hyperparameter_template="better_default",
        compute_oob_performances=False).train(d)
    assert m2.num_trees() == 5
    with pytest.raises(ValueError):
        ydf.GradientBoostedTreesLearner(
            label="LABEL", hyperparameter_template="nope")


def test_best_first_global_growth(tmp_path):
    """Leaf-wise growth (reference growing_strategy=BEST_FIRST_GLOBAL):
    with the same leaf budget it must beat equal-size level-wise trees
    on an interaction target, respect max_num_nodes, and persist."""
    rng = np.random.RandomState(0)
    n = 8000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(2 * x1 - x2 + 0.5 * x1 * x2 > 0, "a", "b")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=25, growing_strategy="BEST_FIRST_GLOBAL",
        max_num_nodes=16, validation_ratio=0.1).train(d)
    assert m.evaluate(d).accuracy > 0.99
    # each tree has at most 16 leaves -> at most 31 nodes
    f = m.forest
    for t in range(f.n_trees):
        lo, hi = f.tree_slice(t)
        assert hi - lo <= 31
    p1 = m.predict(d)
    m.save(str(tmp_path / "bf"))
    np.testing.assert_array_equal(
        p1, ydf.load_model(str(tmp_path / "bf")).predict(d))


def test_best_first_with_subsample():
    """Out-of-sample rows route through the leaf-wise splits replay."""
    rng = np.random.RandomState(1)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(x1 - x2 > 0, "a", "b")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, subsample=0.6,
        growing_strategy="BEST_FIRST_GLOBAL", max_num_nodes=8,
        validation_ratio=0.1).train(d)
    assert m.evaluate(d).accuracy > 0.97


def test_interrupt_returns_partial_model(binary_data, monkeypatch):
    """Fault-injection analogue (reference simulate_worker_failure /
    stop_training_trigger_): an interrupt mid-boosting returns a usable
    partial model instead of crashing."""
    from ydf_amd.learner import trainer as trainer_lib

    orig = trainer_lib.ForestTrainer.grow_tree
    calls = {"n": 0}

    def failing(self, tree_idx, sample_mask=None):
        calls["n"] += 1
        if calls["n"] == 6:
            raise KeyboardInterrupt
        return orig(self, tree_idx, sample_mask)

    monkeypatch.setattr(trainer_lib.ForestTrainer, "grow_tree", failing)
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=50, validation_ratio=0).train(binary_data)
    assert 0 < m.num_trees() <= 5
    assert m.evaluate(binary_data).accuracy > 0.7


def test_crash_recovery_via_snapshot(tmp_path, binary_data, monkeypatch):
    """Crash mid-training (exception after some snapshots), then a new
    learner with resume_training=True continues from the snapshot and
    finishes with the full tree count."""
    from ydf_amd.learner import trainer as trainer_lib

    wd = str(tmp_path / "work")
    kw = dict(label="label", num_trees=30, validation_ratio=0,
              working_dir=wd, resume_training=True,
              resume_training_snapshot_interval_seconds=0.0)

    orig = trainer_lib.ForestTrainer.grow_tree
    calls = {"n": 0}

    def crashing(self, tree_idx, sample_mask=None):
        calls["n"] += 1
        if calls["n"] == 12:
            raise RuntimeError("simulated worker failure")
        return orig(self, tree_idx, sample_mask)

    monkeypatch.setattr(trainer_lib.ForestTrainer, "grow_tree", crashing)
    with pytest.raises(RuntimeError):
        ydf.GradientBoostedTreesLearner(**kw).train(binary_data)
    monkeypatch.setattr(trainer_lib.ForestTrainer, "grow_tree", orig)
    m = ydf.GradientBoostedTreesLearner(**kw).train(binary_data)
    assert m.num_trees() == 30
    assert m.evaluate(binary_data).accuracy > 0.85


def test_appendix_a_hyperparameters():
    """Round-out of the reference's public hyperparameter surface
    (SURVEY Appendix A): l1_regularization shrinks leaves, oblique
    weight families + per-projection feature cap, lambda_loss,
    validation_interval_in_trees, total_max_num_nodes, RF sampling
    without replacement, pure_serving_model, and explicit
    NotImplementedError for LOCAL imputation."""
    d = ydf.generate_synthetic_dataset(num_examples=3000, seed=20)
    m0 = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=15, validation_ratio=0).train(d)
    m1 = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=15, l1_regularization=5.0,
        validation_ratio=0).train(d)
    assert np.abs(m1.forest.thr[m1.forest.feat < 0]).mean() < \
        np.abs(m0.forest.thr[m0.forest.feat < 0]).mean()
    for w in ("POWER_OF_TWO", "INTEGER"):
        mo = ydf.GradientBoostedTreesLearner(
            label="LABEL", num_trees=5, split_axis="SPARSE_OBLIQUE",
            sparse_oblique_weights=w, sparse_oblique_max_num_features=2,
            validation_ratio=0).train(d)
        if len(mo.forest.obl_ranges):
            assert mo.forest.obl_ranges[:, 1].max() <= 2
    mc = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=500, total_max_num_nodes=300,
        validation_ratio=0).train(d)
    assert mc.num_trees() < 500
    mi = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=20, validation_interval_in_trees=5,
        validation_ratio=0.2).train(d)
    assert len(mi.training_logs) == 4
    mr = ydf.RandomForestLearner(
        label="LABEL", num_trees=6, sampling_with_replacement=False,
        bootstrap_size_ratio=0.7,
        compute_oob_performances=False).train(d)
    assert mr.num_trees() == 6
    mp = ydf.GradientBoostedTreesLearner(
        label="LABEL", num_trees=3, pure_serving_model=True,
        validation_ratio=0).train(d)
    assert mp.training_logs is None
    # all three reference missing-value policies are accepted now
    ydf.GradientBoostedTreesLearner(
        label="LABEL", missing_value_policy="RANDOM_LOCAL_IMPUTATION")
    with pytest.raises(ValueError):
        ydf.GradientBoostedTreesLearner(
            label="LABEL", missing_value_policy="NOPE")


def test_edge_cases_robustness():
    """Degenerate inputs train or fail with clear errors (constant
    features, single class, tiny N, all-NaN columns, no features)."""
    m = ydf.GradientBoostedTreesLearner(
        label="l", num_trees=3, validation_ratio=0).train(
        {"x": np.zeros(100, np.float32),
         "y": np.arange(100, dtype=np.float32),
         "l": np.array(["a", "b"] * 50)})
    assert m.evaluate({"x": np.zeros(4, np.float32),
                       "y": np.arange(4, dtype=np.float32),
                       "l": np.array(["a", "b", "a", "b"])}).accuracy >= 0.5
    ydf.GradientBoostedTreesLearner(
        label="l", num_trees=2, validation_ratio=0).train(
        {"x": np.full(60, np.nan, np.float32),
         "z": np.arange(60, dtype=np.float32),
         "l": np.array(["a", "b"] * 30)})
    with pytest.raises(ValueError):
        ydf.GradientBoostedTreesLearner(
            label="l", features=[], num_trees=2,
            validation_ratio=0).train(
            {"x": np.arange(50, dtype=np.float32),
             "l": np.array(["a", "b"] * 25)})


def test_dna_multiclass_real_data():
    """dna.csv (3-class, 180 categorical features): GBT and RF reach the
    reference's published range (~0.94+ accuracy) on a real dataset."""
    import os

    path = ("/root/reference/yggdrasil_decision_forests/test_data/"
            "dataset/dna.csv")
    if not os.path.exists(path):
        pytest.skip("reference test_data not available")
    pd = pytest.importorskip("pandas")
    df = pd.read_csv(path)
    rng = np.random.RandomState(0)
    mask = rng.rand(len(df)) < 0.8
    tr, te = df[mask], df[~mask]
    m = ydf.GradientBoostedTreesLearner(label="LABEL",
                                        num_trees=60).train(tr)
    assert m.evaluate(te).accuracy > 0.92
    mr = ydf.RandomForestLearner(label="LABEL", num_trees=50,
                                 compute_oob_performances=False).train(tr)
    assert mr.evaluate(te).accuracy > 0.90


def test_end_to_end_user_journey(tmp_path):
    """The beginner flow from the reference docs, end to end: train ->
    evaluate -> analyze -> save -> load -> predict -> export (C++ +
    reference format + docker dir) on adult."""
    import os
    import subprocess

    base = "/root/reference/yggdrasil_decision_forests/test_data/dataset"
    if not os.path.exists(base):
        pytest.skip("reference test_data not available")
    pd = pytest.importorskip("pandas")
    train = pd.read_csv(f"{base}/adult_train.csv")
    test = pd.read_csv(f"{base}/adult_test.csv")
    model = ydf.GradientBoostedTreesLearner(
        label="income", num_trees=40).train(train)
    ev = model.evaluate(test)
    assert ev.accuracy > 0.85 and ev.auc > 0.91
    an = model.analyze(test.head(500))
    assert an.variable_importances
    model.save(str(tmp_path / "model"))
    loaded = ydf.load_model(str(tmp_path / "model"))
    np.testing.assert_array_equal(model.predict(test, device="cpu"),
                                  loaded.predict(test, device="cpu"))
    assert "float" in ydf.to_cpp(model)
    ydf.export_ydf_model(model, str(tmp_path / "ydf_format"))
    assert ydf.load_ydf_model(
        str(tmp_path / "ydf_format")).num_trees() == model.num_trees()
    ydf.to_docker(model, str(tmp_path / "serve"))
    assert (tmp_path / "serve" / "Dockerfile").exists()
    assert model.describe("html").startswith("<h2>")


def test_isolation_forest_evaluation_auc():
    """ANOMALY_DETECTION evaluation reports AUC of the anomaly score
    against binary labels (gaussians: reference IF test dataset)."""
    import os

    base = ("/root/reference/yggdrasil_decision_forests/test_data/"
            "dataset")
    if not os.path.exists(base):
        pytest.skip("reference test_data not available")
    pd = pytest.importorskip("pandas")
    tr = pd.read_csv(f"{base}/gaussians_train.csv")
    te = pd.read_csv(f"{base}/gaussians_test.csv")
    m = ydf.IsolationForestLearner(
        label="label",
        features=[c for c in tr.columns if c != "label"]).train(tr)
    assert m.evaluate(te).auc > 0.98


def test_local_imputation_policy():
    """missing_value_policy=LOCAL_IMPUTATION (reference
    decision_tree.proto:85-103): NaN rows ride the reserved bin, the
    scan folds them into the node-local mean bin, splits learn a
    na-direction bit, and serving honors it. When missingness
    correlates with the label tail, LOCAL beats GLOBAL imputation."""
    rng = np.random.RandomState(0)
    n = 8000
    x = rng.randn(n).astype(np.float32)
    miss = rng.rand(n) < 0.35
    y = np.where(np.where(miss, 2.5, x) > 0.5, "hi", "lo")
    xna = x.copy()
    xna[miss] = np.nan
    d = {"x": xna, "z": rng.randn(n).astype(np.float32), "label": y}
    kw = dict(label="label", num_trees=25, validation_ratio=0)
    acc_g = ydf.GradientBoostedTreesLearner(**kw).train(d).evaluate(
        d).accuracy
    ml = ydf.GradientBoostedTreesLearner(
        missing_value_policy="LOCAL_IMPUTATION", **kw).train(d)
    acc_l = ml.evaluate(d).accuracy
    assert ml.forest.na_right.sum() > 0
    assert acc_l >= acc_g
    assert acc_l > 0.995
    # persistence keeps the na bits + policy
    import tempfile

    td = tempfile.mkdtemp()
    ml.save(td)
    m2 = ydf.load_model(td)
    np.testing.assert_array_equal(ml.predict(d), m2.predict(d))
    # RF path too
    mr = ydf.RandomForestLearner(
        label="label", num_trees=10, max_depth=8,
        missing_value_policy="LOCAL_IMPUTATION",
        compute_oob_performances=False).train(d)
    assert mr.evaluate(d).accuracy > 0.99


def test_random_local_imputation():
    """RANDOM_LOCAL_IMPUTATION (reference decision_tree.proto:99-103,
    Random Survival Forests): missing values imputed by sampled
    observed values instead of the mean. On data where the mean sits
    in a low-density region, mean imputation creates a phantom mode
    that hurts; random imputation preserves the distribution."""
    rng = np.random.RandomState(11)
    n = 20000
    # bimodal feature: modes at -2 and +2; mean ~0 is a density valley
    x = np.where(rng.rand(n) < 0.5, -2.0, 2.0) + \
        0.3 * rng.randn(n)
    x = x.astype(np.float32)
    y = np.where(x > 0, "p", "q")
    x_obs = x.copy()
    x_obs[rng.rand(n) < 0.4] = np.nan
    d = {"x": x_obs, "z": rng.randn(n).astype(np.float32), "label": y}

    kw = dict(label="label", num_trees=20, max_depth=4,
              validation_ratio=0.0, device="cpu")
    m_rand = ydf.GradientBoostedTreesLearner(
        missing_value_policy="RANDOM_LOCAL_IMPUTATION", **kw).train(d)
    assert m_rand.evaluate(d).accuracy > 0.75
    # determinism: same seed -> identical model
    m_rand2 = ydf.GradientBoostedTreesLearner(
        missing_value_policy="RANDOM_LOCAL_IMPUTATION", **kw).train(d)
    np.testing.assert_array_equal(m_rand.forest.feat,
                                  m_rand2.forest.feat)
    np.testing.assert_allclose(m_rand.forest.thr, m_rand2.forest.thr)
    # serving is deterministic (global-mean imputation at predict)
    p1 = m_rand.predict(d)
    p2 = m_rand.predict(d)
    np.testing.assert_array_equal(p1, p2)


def test_dart_multiclass_and_na():
    """DART (reference forest_extraction=DART,
    gradient_boosted_trees.h:338) now covers multi-class losses and
    LOCAL_IMPUTATION NA routing: dropout drops whole iterations (all
    class trees together) and dropped-tree replay carries the per-node
    NA direction bits."""
    rng = np.random.RandomState(13)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0.5, "a", np.where(x2 > 0, "b", "c"))
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=30, forest_extraction="DART",
        dart_dropout=0.1, validation_ratio=0.0).train(d)
    ev = m.evaluate(d)
    assert ev.accuracy > 0.95
    p = m.predict(d)
    assert p.shape == (n, 3)
    np.testing.assert_allclose(p.sum(axis=1), 1.0, atol=1e-4)

    # NA routing + DART
    x1na = x1.copy()
    x1na[rng.rand(n) < 0.3] = np.nan
    d2 = {"x1": x1na, "x2": x2,
          "label": np.where(np.nan_to_num(x1na) + x2 > 0, "p", "q")}
    m2 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, forest_extraction="DART",
        dart_dropout=0.1, validation_ratio=0.0,
        missing_value_policy="LOCAL_IMPUTATION").train(d2)
    assert m2.evaluate(d2).accuracy > 0.8


def test_hyperparameter_specification_matches_reference_defaults():
    """Spec generated FROM the signatures (reference
    GetGenericHyperParameterSpecification analogue) pins the Appendix-A
    names/defaults — one source of truth, no drift."""
    from ydf_amd.learner.generic_learner import (
        hyperparameter_specification)

    spec = hyperparameter_specification(ydf.GradientBoostedTreesLearner)
    # reference proto defaults
    pins = {
        "num_trees": 300, "max_depth": 6, "shrinkage": 0.1,
        "subsample": 1.0, "min_examples": 5, "l2_regularization": 0.0,
        "l1_regularization": 0.0, "validation_ratio": 0.1,
        "early_stopping_num_trees_look_ahead": 30,
        "l2_categorical_regularization": 1.0,
        "sparse_oblique_num_projections_exponent": 2.0,
        "sparse_oblique_max_num_projections": 6000,
        "goss_alpha": 0.2, "goss_beta": 0.1,
        "dart_dropout": 0.01, "random_seed": 123456,
        "num_candidate_attributes_ratio": -1.0,
    }
    for k, v in pins.items():
        assert k in spec, f"missing hyperparameter {k}"
        assert spec[k]["default"] == v, (k, spec[k]["default"], v)

    rf = hyperparameter_specification(ydf.RandomForestLearner)
    for k, v in {"num_trees": 300, "bootstrap_size_ratio": 1.0,
                 "winner_take_all": True, "max_depth": 16}.items():
        assert rf[k]["default"] == v, (k, rf[k])

    if_spec = hyperparameter_specification(ydf.IsolationForestLearner)
    assert if_spec["subsample_count"]["default"] == 256


def test_adaptive_work_keeps_tree_count(binary_data):
    """adapt_*_for_maximum_training_duration (reference AdaptativeWork,
    utils/adaptive_work.h:32): the sample shrinks so the forest keeps
    its FULL tree count inside the budget, instead of truncating."""
    import time

    # the assertion is about ADAPTATION (full tree count inside the
    # budget); a contended host (xdist workers) can stall the early
    # full-sample trees past any fixed budget, so retry with a growing
    # one rather than flake
    for budget in (4.0, 15.0, 60.0):
        t0 = time.time()
        m = ydf.RandomForestLearner(
            label="label", num_trees=60, max_depth=10,
            adapt_bootstrap_size_ratio_for_maximum_training_duration=True,
            maximum_training_duration_seconds=budget,
            compute_oob_performances=False, device="cpu").train(
            binary_data)
        took = time.time() - t0
        if m.num_trees() == 60:
            break
    assert m.num_trees() == 60, m.num_trees()
    assert took < budget * 10
    assert m.evaluate(binary_data).accuracy > 0.85

    for budget in (4.0, 15.0, 60.0):
        m2 = ydf.GradientBoostedTreesLearner(
            label="label", num_trees=60, validation_ratio=0.0,
            adapt_subsample_for_maximum_training_duration=True,
            maximum_training_duration_seconds=budget,
            device="cpu").train(binary_data)
        if m2.num_trees() == 60:
            break
    assert m2.num_trees() == 60


def test_analysis_html_report(binary_data):
    """Analysis HTML report with SVG charts (reference
    model_analysis.h CreateHtmlReport + utils/plot.*)."""
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=15,
                                        validation_ratio=0).train(
        binary_data)
    an = m.analyze(binary_data)
    html = an._repr_html_()
    assert "<svg" in html and "polyline" in html  # PDP line plots
    assert "rect" in html                         # VI bar chart
    assert "Partial dependence" in html
    assert html.count("<svg") >= 4  # >=1 VI chart + 3 PDP panels


def test_best_first_categorical():
    """BEST_FIRST_GLOBAL + categorical set-splits (round-1 exclusion
    lifted): leaf-wise growth now scans categorical features host-side
    from the pulled pair histogram."""
    rng = np.random.RandomState(17)
    n = 12000
    cats = rng.randint(0, 10, n)
    probs = np.array([0.9, 0.1, 0.85, 0.15, 0.9, 0.1, 0.8, 0.2,
                      0.88, 0.12])
    y = rng.rand(n) < probs[cats]
    d = {"c": np.array([f"k{v}" for v in cats]),
         "x": rng.randn(n).astype(np.float32),
         "label": np.where(y, "p", "n")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, growing_strategy="BEST_FIRST_GLOBAL",
        max_num_nodes=15, validation_ratio=0.0, device="cpu").train(d)
    assert (m.forest.cat_idx >= 0).any(), "no categorical splits chosen"
    assert m.evaluate(d).accuracy > 0.8
    # round-trip
    import tempfile

    td = tempfile.mkdtemp()
    m.save(td)
    m2 = ydf.load_model(td)
    np.testing.assert_allclose(m.predict(d), m2.predict(d), rtol=1e-5,
                               atol=1e-6)


def test_custom_metrics_in_training_logs(binary_data):
    """User-provided secondary metrics (PYDF custom_metric.py): values
    recorded per validation interval in the training logs."""
    def brier(labels, margins, weights):
        p = 1.0 / (1.0 + np.exp(-margins))
        return float(np.average((p - labels) ** 2, weights=weights))

    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, validation_ratio=0.2,
        custom_metrics=[ydf.BinaryClassificationMetric(
            "brier", brier)]).train(binary_data)
    assert m.training_logs
    vals = [d["brier"] for d in m.training_logs if "brier" in d]
    assert len(vals) >= 10
    assert vals[-1] < vals[0]  # boosting improves it
    assert 0.0 < vals[-1] < 0.25


def test_conditional_expectations(binary_data):
    """CEP plots (reference model_analysis CEP companion to PDP):
    means over actual examples bucketed by feature value."""
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20,
                                        validation_ratio=0).train(
        binary_data)
    an = m.analyze(binary_data)
    assert an.conditional_expectations
    ce = next(c for c in an.conditional_expectations
              if c.feature == "x1")
    assert ce.counts.sum() > 0
    # P(positive class) is monotone in x1 in the fixture (direction
    # depends on the frequency-ordered class vocabulary)
    mp = ce.mean_prediction[ce.counts > 50]
    assert abs(float(mp[-1]) - float(mp[0])) > 0.3
    assert "Conditional expectation" in an._repr_html_()


def test_weight_rescale_semantics_pinned():
    """The automatic heavy-weight rescale (w -> w * 8/max when max > 8;
    packed-u64 histogram field needs per-example h bounded,
    ops/cc/train_kernels.hip) is EXACTLY 'train on the scaled weights':
    pre-scaling the weights by the same factor must give the identical
    model, including with lambda_l2 > 0 where G/(H+lambda) is not
    weight-scale-invariant (documented deviation, ROADMAP.md)."""
    rng = np.random.RandomState(11)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = (x1 + 0.5 * x2 + 0.2 * rng.randn(n)) > 0
    w = rng.uniform(0.5, 50.0, n).astype(np.float32)  # max > 8
    base = {"x1": x1, "x2": x2, "label": np.where(y, "a", "b")}
    kw = dict(label="label", weights="w", num_trees=15, max_depth=4,
              l2_regularization=2.0, validation_ratio=0.0, device="cpu")
    m_auto = ydf.GradientBoostedTreesLearner(**kw).train(
        {**base, "w": w})
    m_pre = ydf.GradientBoostedTreesLearner(**kw).train(
        {**base, "w": w * (8.0 / float(w.max()))})
    np.testing.assert_array_equal(m_auto.forest.feat, m_pre.forest.feat)
    np.testing.assert_allclose(m_auto.forest.thr, m_pre.forest.thr,
                               rtol=1e-6, atol=1e-7)
    np.testing.assert_allclose(m_auto.predict(base),
                               m_pre.predict(base), rtol=1e-6, atol=1e-7)


def test_gbt_reference_param_surface(binary_data):
    """GBT accepts the full reference decision-tree shared parameter
    surface: honest trees, num_candidate_attributes
    (count form), sorting_strategy, keep_non_leaf_label_distribution,
    compute_permutation_variable_importance; unimplemented non-default
    settings raise."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=25, validation_ratio=0.0,
        honest=True, num_candidate_attributes=2,
        sorting_strategy="AUTO").train(binary_data)
    assert m.evaluate(binary_data).accuracy > 0.85

    # honest leaf values differ from plain training (re-estimated on
    # the held-out half) while structure quality stays comparable
    m0 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=25, validation_ratio=0.0).train(
        binary_data)
    assert np.abs(m.predict(binary_data)
                  - m0.predict(binary_data)).max() > 1e-4

    m2 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, validation_ratio=0.0,
        keep_non_leaf_label_distribution=False,
        compute_permutation_variable_importance=True).train(binary_data)
    f = m2.forest
    assert (f.cover[f.feat >= 0] == 0).all()
    assert (f.cover[f.feat < 0] > 0).any()
    vi = m2.variable_importances()
    assert "MEAN_DECREASE_IN_ACCURACY" in vi

    import pytest as _pt
    with _pt.raises(NotImplementedError):
        ydf.GradientBoostedTreesLearner(
            label="label", num_trees=2,
            in_split_min_examples_check=False).train(binary_data)
    with _pt.raises(NotImplementedError):
        ydf.GradientBoostedTreesLearner(
            label="label", num_trees=2,
            mhld_oblique_sample_attributes=True).train(binary_data)


def test_shared_tree_params_accepted_everywhere(binary_data):
    """Every tree learner accepts the reference's shared decision-tree
    parameter surface (PYDF generates all learner signatures from one
    spec): class_weights is honored, num_discretized_numerical_bins
    feeds binning, unimplemented settings raise, unknown kwargs are
    rejected."""
    # class_weights shifts the decision boundary toward the upweighted
    # class -> more "yes" predictions
    m0 = ydf.RandomForestLearner(
        label="label", num_trees=15,
        compute_oob_performances=False).train(binary_data)
    m1 = ydf.RandomForestLearner(
        label="label", num_trees=15, compute_oob_performances=False,
        class_weights={"yes": 8.0}).train(binary_data)
    r0 = (m0.predict(binary_data) > 0.5).mean()
    r1 = (m1.predict(binary_data) > 0.5).mean()
    assert r1 > r0 + 0.02, (r0, r1)

    # coarser discretization trains and degrades gracefully
    m2 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, validation_ratio=0.0,
        num_discretized_numerical_bins=16).train(binary_data)
    assert m2.evaluate(binary_data).accuracy > 0.85

    # accepted names on every learner (no TypeError at construction)
    for cls in (ydf.CartLearner, ydf.IsolationForestLearner):
        cls(label="label", sorting_strategy="AUTO",
            sparse_oblique_weights_integer_minimum=-4,
            numerical_vector_sequence_num_random_anchors=10)

    import pytest as _pt
    with _pt.raises(TypeError):
        ydf.RandomForestLearner(label="label", not_a_param=1)
    with _pt.raises(NotImplementedError):
        ydf.CartLearner(label="label",
                        include_all_columns=True).train(binary_data)


def test_data_spec_override_and_dgbt(binary_data):
    """Learner(data_spec=...) trains against a pre-built dataspec
    (reference generic data_spec arg), and the distributed GBT class
    exposes the reference worker-pool surface with RCCL guidance."""
    m0 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, validation_ratio=0.0).train(
        binary_data)
    m1 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, validation_ratio=0.0,
        data_spec=m0.data_spec()).train(binary_data)
    np.testing.assert_array_equal(m0.forest.feat, m1.forest.feat)
    np.testing.assert_allclose(m0.predict(binary_data),
                               m1.predict(binary_data), rtol=1e-6)

    m2 = ydf.DistributedGradientBoostedTreesLearner(
        label="label", num_trees=10, validation_ratio=0.0,
        worker_logs=False, force_numerical_discretization=True).train(
        binary_data)
    assert m2.evaluate(binary_data).accuracy > 0.9
    with pytest.raises(NotImplementedError, match="torch.distributed"):
        ydf.DistributedGradientBoostedTreesLearner(
            label="label", workers=["host:2001"])


def test_predict_shap_reference_tuple_form(regression_data):
    """predict_shap unpacks as (values, initial_value) like the
    reference (generic_model.py:507) while the legacy mapping form
    keeps working."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=10,
        validation_ratio=0).train(regression_data)
    values, initial = m.predict_shap(regression_data, num_threads=2)
    assert set(values) == set(m.input_feature_names())
    total = np.stack(list(values.values()), 1).sum(1) + initial
    np.testing.assert_allclose(total, m.predict(regression_data),
                               atol=1e-4)
