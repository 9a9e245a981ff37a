"""Exact (non-binned) split oracle tests.

Reference analogue: the default exact numerical splitter
(splitter_scanner.h:933-1101 ScanSplits / :1290 presorted scan) and
TrainAndTestTester metric-margin golden checks (utils/test_utils.h:79,
YDF_TEST_METRIC :428). Structural golden equality requires reproducing
the reference's RNG consumption and is out of scope (the reference
itself gates it off by default, test_utils.cc:1107); these tests pin
(a) exactness against brute force, (b) a case binning provably loses,
(c) metric-margin agreement with the reference's own golden models.
"""
import os

import numpy as np
import pytest

import ydf_amd as ydf
from ydf_amd.learner.exact import ExactSplitter

REF = "/root/reference/yggdrasil_decision_forests/test_data"


def test_root_split_matches_brute_force():
    rng = np.random.RandomState(7)
    N, F = 500, 4
    X = rng.randn(F, N).astype(np.float32)
    g = rng.randn(N)
    h = np.abs(rng.randn(N)) + 0.1
    sp = ExactSplitter(X, None, max_depth=1, min_examples=1,
                       min_hessian=0.0, lambda_l2=0.5)
    t = sp.grow_tree(g, h)
    G, H = g.sum(), h.sum()
    best = -np.inf
    bf = -1
    for f in range(F):
        o = np.argsort(X[f])
        cg, ch = np.cumsum(g[o]), np.cumsum(h[o])
        sv = X[f][o]
        for i in range(N - 1):
            if sv[i] == sv[i + 1]:
                continue
            GL, HL = cg[i], ch[i]
            GR, HR = G - GL, H - HL
            gain = GL * GL / (HL + 0.5) + GR * GR / (HR + 0.5) \
                - G * G / (H + 0.5)
            if gain > best:
                best, bf = gain, f
    assert t.feat[0] == bf
    np.testing.assert_allclose(t.gain[0], best, rtol=1e-5)


def test_exact_beats_binning_on_adversarial_data():
    """Feature where the informative boundary sits INSIDE one quantile
    bin: 256-bin training cannot split there, exact training can."""
    rng = np.random.RandomState(1)
    n = 50000
    # 99.8% of mass in [0,1); 0.2% carries the label boundary at 100.5.
    # 0.2% < 1/256, so no quantile cut can land inside the tail: the
    # whole tail shares one bin and binned training cannot split it.
    x = rng.rand(n).astype(np.float32)
    hot = rng.rand(n) < 0.002
    x[hot] = 100.0 + rng.rand(hot.sum()).astype(np.float32)
    y_bool = np.where(hot, x > 100.5, rng.rand(n) < 0.5)
    data = {"x": x, "label": np.where(y_bool, "a", "b")}

    kw = dict(label="label", num_trees=20, max_depth=4,
              validation_ratio=0.0, device="cpu")
    m_binned = ydf.GradientBoostedTreesLearner(**kw).train(data)
    m_exact = ydf.GradientBoostedTreesLearner(
        discretize_numerical_columns=False, **kw).train(data)

    hot_data = {k: np.asarray(v)[hot] for k, v in data.items()}
    acc_b = m_binned.evaluate(hot_data).accuracy
    acc_e = m_exact.evaluate(hot_data).accuracy
    assert acc_e > 0.99, acc_e
    assert acc_e > acc_b + 0.1, (acc_e, acc_b)


def test_exact_matches_binned_when_binning_lossless():
    """<=255 distinct values per feature: quantile bins are lossless, so
    exact and binned training must produce equally good models."""
    rng = np.random.RandomState(2)
    n = 8000
    x1 = rng.randint(0, 100, n).astype(np.float32)
    x2 = rng.randint(0, 50, n).astype(np.float32)
    y = np.where((2 * x1 - x2 + 20 * rng.randn(n)) > 75, "y", "n")
    data = {"x1": x1, "x2": x2, "label": y}
    kw = dict(label="label", num_trees=30, validation_ratio=0.0,
              device="cpu")
    acc_b = ydf.GradientBoostedTreesLearner(**kw).train(data) \
        .evaluate(data).accuracy
    acc_e = ydf.GradientBoostedTreesLearner(
        discretize_numerical_columns=False, **kw).train(data) \
        .evaluate(data).accuracy
    assert abs(acc_e - acc_b) < 0.01, (acc_e, acc_b)


@pytest.mark.skipif(not os.path.exists(REF),
                    reason="reference test_data not available")
def test_golden_margin_adult_classification():
    """Margin-pinned agreement with the reference golden model
    gbt_adult_base: our exact-split GBT trained on adult_train with the
    reference default shape must reach the golden model's test accuracy
    within the TrainAndTestTester-style margin."""
    pd = pytest.importorskip("pandas")
    tr = pd.read_csv(f"{REF}/dataset/adult_train.csv")
    te = pd.read_csv(f"{REF}/dataset/adult_test.csv")
    golden = ydf.load_ydf_model(f"{REF}/model/adult_binary_class_gbdt")
    gp = golden.predict(te, device="cpu")
    y = (te["income"].values == ">50K")
    acc_golden = ((gp > 0.5) == y).mean()

    m = ydf.GradientBoostedTreesLearner(
        label="income", num_trees=100, max_depth=6,
        validation_ratio=0.0, discretize_numerical_columns=False,
        device="cpu").train(tr)
    p = m.predict(te, device="cpu")
    acc = ((p > 0.5) == y).mean()
    # YDF_TEST_METRIC-style margin (reference adult tests use ~0.01)
    assert acc > acc_golden - 0.012, (acc, acc_golden)


@pytest.mark.skipif(not os.path.exists(REF),
                    reason="reference test_data not available")
def test_golden_margin_abalone_regression():
    """Same margin pin against gbt_abalone (regression RMSE)."""
    pd = pytest.importorskip("pandas")
    df = pd.read_csv(f"{REF}/dataset/abalone.csv")
    golden = ydf.load_ydf_model(f"{REF}/model/abalone_regression_gbdt")
    gp = golden.predict(df, device="cpu")
    y = df["Rings"].values.astype(np.float64)
    rmse_golden = float(np.sqrt(np.mean((gp - y) ** 2)))

    m = ydf.GradientBoostedTreesLearner(
        label="Rings", task=ydf.Task.REGRESSION, num_trees=100,
        max_depth=6, validation_ratio=0.0,
        discretize_numerical_columns=False, device="cpu").train(df)
    p = m.predict(df, device="cpu")
    rmse = float(np.sqrt(np.mean((p - y) ** 2)))
    # training-set RMSE of a fresh 100-tree model must at least match
    # the golden model's (which saw a train/test split)
    assert rmse < rmse_golden * 1.10, (rmse, rmse_golden)


def test_mhld_oblique_beats_axis_aligned_on_rotated_data():
    """MHLD oblique (reference oblique.h:33, greedy LDA subsets): on a
    rotated decision boundary the LDA projection recovers the
    separating direction that axis-aligned stumps cannot express."""
    rng = np.random.RandomState(21)
    n = 6000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    # boundary along x1 + x2 (45 degrees)
    y = np.where(x1 + x2 + 0.15 * rng.randn(n) > 0, "p", "q")
    d = {"x1": x1, "x2": x2, "label": y}
    kw = dict(label="label", num_trees=4, max_depth=2,
              validation_ratio=0.0, device="cpu")
    m_axis = ydf.GradientBoostedTreesLearner(**kw).train(d)
    m_mhld = ydf.GradientBoostedTreesLearner(
        split_axis="MHLD_OBLIQUE", **kw).train(d)
    acc_a = m_axis.evaluate(d).accuracy
    acc_m = m_mhld.evaluate(d).accuracy
    assert acc_m > 0.95, acc_m
    assert acc_m > acc_a + 0.03, (acc_m, acc_a)
    # oblique conditions present + model round-trips
    assert len(m_mhld.forest.obl_ranges) > 0
    p1 = m_mhld.predict(d)
    import tempfile

    td = tempfile.mkdtemp()
    m_mhld.save(td)
    import ydf_amd

    m2 = ydf_amd.load_model(td)
    np.testing.assert_allclose(p1, m2.predict(d), rtol=1e-5, atol=1e-6)
