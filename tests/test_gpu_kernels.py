"""HIP kernel numerics tests: every GPU op against its C++ CPU twin and/or a
plain torch fp32 reference. Run on an MI355X box (`pytest -m gpu`)."""
import numpy as np
import pytest
import torch

import ydf_amd as ydf
from ydf_amd import ops

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="no GPU")


@pytest.fixture(scope="module", autouse=True)
def _check_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU available")


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda:0")


def test_native_extension_loaded():
    import ydf_amd._ydf_ops as m

    assert "ydf_amd" in m.__file__, m.__file__


def test_bin_data_gpu_vs_cpu(dev):
    rng = np.random.RandomState(0)
    x = rng.randn(5, 100000).astype(np.float32)
    bnd = np.sort(rng.randn(5, 255).astype(np.float32), axis=1)
    xt = torch.from_numpy(x)
    bt = torch.from_numpy(bnd)
    out_cpu = torch.empty(xt.shape, dtype=torch.uint8)
    ops.bin_data(xt, bt, out_cpu)
    out_gpu = torch.empty(xt.shape, dtype=torch.uint8, device=dev)
    ops.bin_data(xt.to(dev), bt.to(dev), out_gpu)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(out_gpu.cpu().numpy(), out_cpu.numpy())


def test_grad_hess_gpu_vs_torch(dev):
    rng = np.random.RandomState(1)
    preds = torch.from_numpy(rng.randn(50000).astype(np.float32)).to(dev)
    labels = torch.from_numpy(
        (rng.rand(50000) > 0.5).astype(np.float32)).to(dev)
    gh = torch.empty((50000, 2), dtype=torch.float32, device=dev)
    ops.grad_hess(preds, labels, gh, 1)
    torch.cuda.synchronize()
    p = torch.sigmoid(preds)
    torch.testing.assert_close(gh[:, 0], p - labels, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(gh[:, 1], (p * (1 - p)).clamp_min(1e-16),
                               rtol=1e-4, atol=1e-5)


def test_hist_build_gpu_vs_cpu_exact_integers(dev):
    """Integer-valued {g,h} make float sums order-independent, so GPU LDS
    atomics must equal CPU sequential sums EXACTLY."""
    rng = np.random.RandomState(2)
    F, N = 6, 200000
    bins = rng.randint(0, 256, size=(F, N)).astype(np.uint8)
    g = rng.randint(-8, 8, N).astype(np.float32)
    h = rng.randint(0, 4, N).astype(np.float32)
    node_ids = (rng.randint(0, 8, N) + 7).astype(np.int32)
    gh = torch.from_numpy(np.stack([g, h], 1).copy())
    slot_map = torch.arange(8, dtype=torch.int32)
    hist_c = torch.zeros((8, F, 256, 3))
    ops.hist_build(torch.from_numpy(bins), gh, torch.from_numpy(node_ids),
                   slot_map, hist_c, 7, 8, 0, 8)
    hist_g = torch.zeros((8, F, 256, 3), device=dev)
    ops.hist_build(torch.from_numpy(bins).to(dev), gh.to(dev),
                   torch.from_numpy(node_ids).to(dev), slot_map.to(dev),
                   hist_g, 7, 8, 0, 8)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(hist_g.cpu().numpy(), hist_c.numpy())


def test_split_scan_gpu_vs_cpu(dev):
    rng = np.random.RandomState(3)
    F, B, n_slots = 7, 256, 5
    hist = rng.randint(0, 50, size=(n_slots, F, B, 3)).astype(np.float32)
    ht = torch.from_numpy(hist)
    args = dict(lambda_l2=1.0, min_hessian=0.0, min_examples=5, min_gain=0.0)
    abs_of_slot = torch.arange(7, 7 + n_slots, dtype=torch.int32)

    def run(device):
        h = ht.to(device)
        aos = abs_of_slot.to(device)
        ns = torch.zeros((63, 3), device=device)
        bg = torch.empty((n_slots, F), device=device)
        bb = torch.empty((n_slots, F), dtype=torch.int32, device=device)
        bf = torch.empty(n_slots, dtype=torch.int32, device=device)
        bbin = torch.empty(n_slots, dtype=torch.int32, device=device)
        bgain = torch.empty(n_slots, device=device)
        ops.split_scan(h, aos, ns, bg, bb, bf, bbin, bgain, 0, n_slots,
                       args["lambda_l2"], args["min_hessian"],
                       args["min_examples"], args["min_gain"])
        if device != "cpu":
            torch.cuda.synchronize()
        return (bf.cpu().numpy(), bbin.cpu().numpy(), bgain.cpu().numpy(),
                ns.cpu().numpy())

    f_c, b_c, g_c, ns_c = run("cpu")
    f_g, b_g, g_g, ns_g = run(dev)
    np.testing.assert_array_equal(f_g, f_c)
    np.testing.assert_array_equal(b_g, b_c)
    np.testing.assert_allclose(g_g, g_c, rtol=1e-4)
    np.testing.assert_allclose(ns_g, ns_c, rtol=1e-4, atol=1e-3)


def test_update_node_ids_gpu_vs_cpu(dev):
    rng = np.random.RandomState(4)
    F, N = 4, 300000
    bins = rng.randint(0, 256, size=(F, N)).astype(np.uint8)
    node_ids0 = (rng.randint(0, 4) * 0 + rng.randint(3, 7, N)).astype(
        np.int32)
    best_feat = rng.randint(-1, F, 4).astype(np.int32)
    best_bin = rng.randint(0, 255, 4).astype(np.int32)
    slot_map = torch.arange(4, dtype=torch.int32)
    ids_c = torch.from_numpy(node_ids0.copy())
    ops.update_node_ids(torch.from_numpy(bins), ids_c, slot_map,
                        torch.from_numpy(best_feat),
                        torch.from_numpy(best_bin), 3, 4)
    ids_g = torch.from_numpy(node_ids0.copy()).to(dev)
    ops.update_node_ids(torch.from_numpy(bins).to(dev), ids_g,
                        slot_map.to(dev),
                        torch.from_numpy(best_feat).to(dev),
                        torch.from_numpy(best_bin).to(dev), 3, 4)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(ids_g.cpu().numpy(), ids_c.numpy())


def test_predict_forest_gpu_vs_cpu(dev, binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=30,
                                        device="cpu").train(binary_data)
    p_cpu = m.predict(binary_data, device="cpu")
    p_gpu = m.predict(binary_data, device="cuda")
    np.testing.assert_allclose(p_gpu, p_cpu, rtol=1e-5, atol=1e-6)


def test_rf_gpu_bit_exact_vs_cpu(binary_data):
    """Integer gradient sums: the GPU-trained forest must match the
    CPU-trained forest exactly (same argmax, same thresholds)."""
    kw = dict(label="label", num_trees=3, max_depth=6,
              bootstrap_training_dataset=False, num_candidate_attributes=-1)
    m_cpu = ydf.RandomForestLearner(device="cpu", **kw).train(binary_data)
    m_gpu = ydf.RandomForestLearner(device="cuda", **kw).train(binary_data)
    np.testing.assert_array_equal(m_gpu.forest.feat, m_cpu.forest.feat)
    np.testing.assert_array_equal(m_gpu.forest.left, m_cpu.forest.left)
    np.testing.assert_allclose(m_gpu.forest.thr, m_cpu.forest.thr, rtol=1e-6)


def test_gbt_gpu_quality(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=100,
                                        device="cuda").train(binary_data)
    ev = m.evaluate(binary_data, device="cuda")
    assert ev.accuracy > 0.93
    assert ev.auc > 0.97


def test_gbt_gpu_regression(regression_data):
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=100,
        device="cuda").train(regression_data)
    ev = m.evaluate(regression_data, device="cuda")
    assert ev.rmse < 0.5


def test_gbt_gpu_multiclass():
    rng = np.random.RandomState(2)
    n = 20000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0.5, "a", np.where(x2 > 0, "b", "c"))
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=40,
                                        device="cuda").train(d)
    assert m.evaluate(d, device="cuda").accuracy > 0.98


def test_rf_gpu_feature_sampling_and_bootstrap(binary_data):
    m = ydf.RandomForestLearner(label="label", num_trees=20, max_depth=10,
                                device="cuda").train(binary_data)
    assert m.evaluate(binary_data, device="cuda").accuracy > 0.9


def test_binary_logloss_gpu(dev):
    rng = np.random.RandomState(0)
    preds = torch.from_numpy(rng.randn(100000).astype(np.float32)).to(dev)
    labels = torch.from_numpy(
        (rng.rand(100000) > 0.4).astype(np.float32)).to(dev)
    out = torch.zeros(2, device=dev)
    ops.binary_logloss(preds, labels, out)
    torch.cuda.synchronize()
    ref = torch.nn.functional.binary_cross_entropy_with_logits(
        preds, labels, reduction="sum")
    np.testing.assert_allclose(out[0].item(), ref.item(), rtol=1e-3)


def test_categorical_gpu_vs_cpu():
    """Categorical set-splits: GPU bitonic ordering must equal CPU argsort
    (integer-valued stats keep everything exact)."""
    rng = np.random.RandomState(11)
    n = 30000
    cats = rng.randint(0, 12, n)
    y = (cats % 3 == 0)
    d = {"c": np.array([f"cat{v}" for v in cats]),
         "x": rng.randint(-5, 5, n).astype(np.float32),
         "label": np.where(y, "p", "n")}
    kw = dict(label="label", num_trees=3, max_depth=4,
              bootstrap_training_dataset=False, num_candidate_attributes=-1)
    m_cpu = ydf.RandomForestLearner(device="cpu", **kw).train(d)
    m_gpu = ydf.RandomForestLearner(device="cuda", **kw).train(d)
    np.testing.assert_array_equal(m_gpu.forest.feat, m_cpu.forest.feat)
    np.testing.assert_array_equal(m_gpu.forest.cat_idx, m_cpu.forest.cat_idx)
    np.testing.assert_array_equal(m_gpu.forest.masks, m_cpu.forest.masks)
    np.testing.assert_allclose(m_gpu.predict(d, device="cuda"),
                               m_cpu.predict(d, device="cpu"), rtol=1e-5,
                               atol=1e-6)


def test_adult_gpu_quality(adult_paths):
    pd = pytest.importorskip("pandas")
    tr, te = adult_paths
    m = ydf.GradientBoostedTreesLearner(label="income",
                                        device="cuda").train(pd.read_csv(tr))
    ev = m.evaluate(pd.read_csv(te), device="cuda")
    assert ev.accuracy > 0.86
    assert ev.auc > 0.92


def test_monotonic_gpu(regression_data):
    feats = [ydf.Feature("x1", monotonic=1), ydf.Feature("x2")]
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, features=feats,
        num_trees=40, validation_ratio=0, device="cuda").train(
            regression_data)
    grid = np.linspace(-3, 3, 200).astype(np.float32)
    p = m.predict({"x1": grid, "x2": np.zeros_like(grid)}, device="cuda")
    assert np.diff(p).min() >= -1e-6


def test_oblique_predict_gpu_vs_cpu():
    """Hand-built oblique forest: GPU inference kernel vs CPU twin
    (both LDS and global-memory fallback paths exercise the oblique
    branch; no reference data needed)."""
    rng = np.random.RandomState(5)
    F, N = 6, 50000
    X = rng.randn(F, N).astype(np.float32)
    from ydf_amd.model.forest import FlatForest

    # tree: oblique root (0.5*x0 - x2 + 2*x4 > 0.1); children split on
    # x1 (no branch) / x3 (yes branch); leaves 1..4
    forest = FlatForest(
        feat=np.array([0, 1, 3, -1, -1, -1, -1], np.int32),
        thr=np.array([0.1, 0.0, -0.3, 1.0, 2.0, 3.0, 4.0], np.float32),
        left=np.array([1, 3, 5, 0, 0, 0, 0], np.int32),
        roots=np.array([0], np.int32),
        cat_idx=np.array([-2, -1, -1, -1, -1, -1, -1], np.int32),
        obl_ranges=np.array([[0, 3]], np.int32),
        obl_attr=np.array([0, 2, 4], np.int32),
        obl_w=np.array([0.5, -1.0, 2.0], np.float32),
    )
    from ydf_amd.model.generic_model import _DeviceForest
    out = {}
    for dev in ("cpu", "cuda"):
        d = torch.device(dev)
        df = _DeviceForest(forest, d)
        Xd = torch.from_numpy(X).to(d)
        o = torch.empty(N, dtype=torch.float32, device=d)
        ops.predict_forest(Xd, df.feat, df.thr, df.left, df.roots, o,
                           cat_idx=df.cat_idx, masks=df.masks,
                           packed=df.packed, obl_ranges=df.obl_ranges,
                           obl_attr=df.obl_attr, obl_w=df.obl_w)
        out[dev] = o.cpu().numpy()
    # reference semantics in numpy
    dot = 0.5 * X[0] - X[2] + 2.0 * X[4]
    right = dot > 0.1
    lv = np.where(right, np.where(X[3] > -0.3, 4.0, 3.0),
                  np.where(X[1] > 0.0, 2.0, 1.0)).astype(np.float32)
    np.testing.assert_allclose(out["cpu"], lv, atol=1e-6)
    np.testing.assert_allclose(out["cuda"], lv, atol=1e-6)


def test_oblique_gbt_train_gpu():
    """Oblique training end-to-end on the GPU (GEMM projections +
    virtual-feature histograms), diagonal boundary."""
    rng = np.random.RandomState(0)
    n = 20000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    tr = {"x1": x1, "x2": x2, "label": np.where(x1 + x2 > 0, "p", "n")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, max_depth=3, validation_ratio=0.1,
        split_axis="SPARSE_OBLIQUE", device="cuda").train(tr)
    assert (m.forest.cat_idx <= -2).sum() > 0
    t1 = rng.randn(4000).astype(np.float32)
    t2 = rng.randn(4000).astype(np.float32)
    te = {"x1": t1, "x2": t2, "label": np.where(t1 + t2 > 0, "p", "n")}
    assert m.evaluate(te, device="cuda").accuracy > 0.99


def test_rf_deep_partitioned_i16_vs_cpu():
    """Deep levels route through the feature-interleaved gathered
    kernel (hist_build_gathered16); integer-valued RF gradients make the
    GPU forest bit-comparable to the CPU sparse path."""
    rng = np.random.RandomState(11)
    n, F = 60000, 40
    d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(F)}
    d["label"] = np.where(
        rng.randn(n) + d["x0"] * 2 - d["x1"] + d["x2"] * d["x3"] > 0,
        "a", "b")
    kw = dict(label="label", num_trees=3, max_depth=12, min_examples=2,
              bootstrap_training_dataset=False,
              num_candidate_attributes=-1,
              compute_oob_performances=False)
    m_cpu = ydf.RandomForestLearner(device="cpu", **kw).train(d)
    m_gpu = ydf.RandomForestLearner(device="cuda", **kw).train(d)
    np.testing.assert_array_equal(m_gpu.forest.feat, m_cpu.forest.feat)
    np.testing.assert_array_equal(m_gpu.forest.left, m_cpu.forest.left)
    np.testing.assert_allclose(m_gpu.forest.thr, m_cpu.forest.thr,
                               rtol=1e-5, atol=1e-6)


def test_rf_deep_partitioned_i16_masked_vs_unmasked():
    """Masked interleaved build (per-slot feature-sampling bits) must
    reproduce the unmasked per-feature gathered path exactly: identical
    masks (same device generator), and integer-valued RF gradient sums
    make histogram subtraction (active only in the unmasked path) exact.
    CPU cannot be the reference here: feature masks are drawn with the
    device's own generator."""
    import os

    rng = np.random.RandomState(12)
    n, F = 60000, 40
    d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(F)}
    d["label"] = np.where(
        rng.randn(n) + d["x0"] * 2 - d["x1"] + d["x2"] * d["x3"] > 0,
        "a", "b")
    kw = dict(label="label", num_trees=3, max_depth=12, min_examples=2,
              bootstrap_training_dataset=False,
              num_candidate_attributes=6,
              compute_oob_performances=False, device="cuda")
    os.environ["YDFA_HIST_I16"] = "0"
    try:
        m_ref = ydf.RandomForestLearner(**kw).train(d)
    finally:
        os.environ.pop("YDFA_HIST_I16")
    m_i16 = ydf.RandomForestLearner(**kw).train(d)
    np.testing.assert_array_equal(m_i16.forest.feat, m_ref.forest.feat)
    np.testing.assert_array_equal(m_i16.forest.left, m_ref.forest.left)
    np.testing.assert_allclose(m_i16.forest.thr, m_ref.forest.thr,
                               rtol=1e-5, atol=1e-6)


def test_best_first_gpu():
    """Leaf-wise growth on the GPU (single-node hist/scan launches)."""
    rng = np.random.RandomState(13)
    n = 50000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    d = {"x1": x1, "x2": x2,
         "label": np.where(2 * x1 - x2 + 0.5 * x1 * x2 > 0, "a", "b")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=15, growing_strategy="BEST_FIRST_GLOBAL",
        max_num_nodes=16, validation_ratio=0.1, device="cuda").train(d)
    assert m.evaluate(d, device="cuda").accuracy > 0.98


def test_cox_gpu():
    rng = np.random.RandomState(14)
    n = 30000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    hazard = np.exp(x1 - 0.5 * x2)
    T = rng.exponential(1.0 / hazard)
    C = rng.exponential(2.0, n)
    d = {"x1": x1, "x2": x2,
         "time": np.minimum(T, C).astype(np.float32),
         "event": T <= C}
    m = ydf.GradientBoostedTreesLearner(
        label="time", label_event_observed="event",
        task=ydf.Task.SURVIVAL_ANALYSIS, num_trees=30,
        validation_ratio=0.1, device="cuda").train(d)
    assert m.evaluate(d).cindex > 0.7


def test_quickscorer_matches_flat_kernel():
    """HIP QuickScorer engine (reference quick_scorer_extended) must
    reproduce the flat-node kernel exactly on a numerical GBT."""
    from ydf_amd.model.forest import build_quickscorer

    d = ydf.generate_synthetic_dataset(num_examples=20000,
                                       num_numerical=8,
                                       num_categorical=0, num_boolean=0,
                                       seed=9)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=50,
                                        validation_ratio=0,
                                        device="cuda").train(d)
    conds, offs, lv = build_quickscorer(m.forest)
    X = torch.from_numpy(np.ascontiguousarray(
        m._encode_features(d))).cuda()
    want = m.predict_margin(X)[0]
    out = torch.empty(X.shape[1], dtype=torch.float32, device="cuda")
    ops.predict_forest_qs(
        X, torch.from_numpy(conds).cuda(),
        torch.from_numpy(offs).cuda(), torch.from_numpy(lv).cuda(), out,
        init=float(m.init_predictions[0]), scale=m._leaf_scale())
    np.testing.assert_allclose(out.cpu().numpy(), want.cpu().numpy(),
                               rtol=1e-5, atol=1e-5)


def test_8bit_binned_engine_matches_flat():
    """8-bit serving engine (pre-binned u8 features, bin-index
    thresholds) must match the float flat-node kernel on our own
    models (thresholds sit exactly on training cuts)."""
    from ydf_amd.model.forest import (pack_binned_nodes,
                                      padded_boundaries)

    d = ydf.generate_synthetic_dataset(num_examples=30000,
                                       num_numerical=10,
                                       num_categorical=0, num_boolean=0,
                                       seed=21)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=40,
                                        validation_ratio=0,
                                        device="cuda").train(d)
    bnd = padded_boundaries(m.dataspec.feature_columns)
    X = torch.from_numpy(np.ascontiguousarray(
        m._encode_features(d))).cuda()
    want = m.predict_margin(X)[0]
    bins = torch.empty(X.shape, dtype=torch.uint8, device="cuda")
    ops.bin_data(X, torch.from_numpy(bnd).cuda(), bins)
    packed = torch.from_numpy(pack_binned_nodes(m.forest, bnd)).cuda()
    roots = torch.from_numpy(m.forest.roots).cuda()
    out = torch.empty(X.shape[1], dtype=torch.float32, device="cuda")
    ops.predict_forest_binned(bins, packed, roots, out,
                              init=float(m.init_predictions[0]),
                              scale=m._leaf_scale())
    np.testing.assert_allclose(out.cpu().numpy(), want.cpu().numpy(),
                               rtol=1e-5, atol=1e-5)


def test_force_engine_equality_gpu():
    """model.force_engine('qs'/'8bit'): engine-routed predictions match
    the default flat kernel (reference engine cross-check discipline)."""
    d = ydf.generate_synthetic_dataset(num_examples=20000,
                                       num_numerical=6,
                                       num_categorical=0, num_boolean=0,
                                       seed=22)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=30,
                                        validation_ratio=0,
                                        device="cuda").train(d)
    want = m.predict(d, device="cuda")
    for eng in m.list_compatible_engines():
        m.force_engine(eng)
        got = m.predict(d, device="cuda")
        np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-5,
                                   err_msg=eng)
    m.force_engine(None)


def test_na_routing_gpu_vs_cpu():
    """na_value routing (imported reference models): NaN inputs follow
    the stored per-node direction identically on CPU and GPU."""
    from ydf_amd.model.forest import FlatForest
    from ydf_amd.model.generic_model import _DeviceForest

    rng = np.random.RandomState(6)
    F, N = 4, 40000
    X = rng.randn(F, N).astype(np.float32)
    X[0, rng.rand(N) < 0.3] = np.nan
    X[1, rng.rand(N) < 0.3] = np.nan
    forest = FlatForest(
        feat=np.array([0, 1, 1, -1, -1, -1, -1], np.int32),
        thr=np.array([0.2, -0.1, 0.4, 1.0, 2.0, 3.0, 4.0], np.float32),
        left=np.array([1, 3, 5, 0, 0, 0, 0], np.int32),
        roots=np.array([0], np.int32),
        na_right=np.array([1, 0, 1, 0, 0, 0, 0], np.uint8))
    outs = {}
    for dev in ("cpu", "cuda"):
        d = torch.device(dev)
        df = _DeviceForest(forest, d)
        Xd = torch.from_numpy(X).to(d)
        o = torch.empty(N, dtype=torch.float32, device=d)
        ops.predict_forest(Xd, df.feat, df.thr, df.left, df.roots, o,
                           cat_idx=df.cat_idx, masks=df.masks,
                           packed=df.packed, na_right=df.na_right)
        outs[dev] = o.cpu().numpy()
    # reference semantics in numpy
    r0 = np.where(np.isnan(X[0]), 1, (X[0] > 0.2).astype(int))
    rl = np.where(np.isnan(X[1]), 0, (X[1] > -0.1).astype(int))
    rr = np.where(np.isnan(X[1]), 1, (X[1] > 0.4).astype(int))
    want = np.where(r0 == 1, np.where(rr == 1, 4.0, 3.0),
                    np.where(rl == 1, 2.0, 1.0)).astype(np.float32)
    np.testing.assert_allclose(outs["cpu"], want, atol=1e-6)
    np.testing.assert_allclose(outs["cuda"], want, atol=1e-6)


def test_deep_mlp_gpu():
    """Deep learners train on the GPU through torch (rocBLAS/MIOpen)."""
    d = ydf.generate_synthetic_dataset(num_examples=4000,
                                       num_numerical=5,
                                       num_categorical=1, seed=30)
    m = ydf.MultiLayerPerceptronLearner(
        label="LABEL", num_epochs=15, num_layers=2, layer_size=48,
        device="cuda").train(d)
    assert m.evaluate(d, device="cuda").auc > 0.75


def test_local_imputation_gpu_vs_cpu():
    """LOCAL_IMPUTATION forests are bit-comparable across devices (RF
    integer gradients; same na-bin merge on both paths)."""
    rng = np.random.RandomState(31)
    n = 40000
    x = rng.randn(n).astype(np.float32)
    miss = rng.rand(n) < 0.3
    xna = x.copy()
    xna[miss] = np.nan
    d = {"x": xna, "z": rng.randn(n).astype(np.float32),
         "label": np.where(np.where(miss, 2.0, x) > 0.4, "a", "b")}
    kw = dict(label="label", num_trees=3, max_depth=8,
              bootstrap_training_dataset=False,
              num_candidate_attributes=-1,
              missing_value_policy="LOCAL_IMPUTATION",
              compute_oob_performances=False)
    m_cpu = ydf.RandomForestLearner(device="cpu", **kw).train(d)
    m_gpu = ydf.RandomForestLearner(device="cuda", **kw).train(d)
    np.testing.assert_array_equal(m_gpu.forest.feat, m_cpu.forest.feat)
    np.testing.assert_array_equal(m_gpu.forest.na_right,
                                  m_cpu.forest.na_right)
    np.testing.assert_allclose(m_gpu.forest.thr, m_cpu.forest.thr,
                               rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_binned8_engine_matches_flat(binary_data):
    """Compact 8-byte-node binned engine == flat engine on a trained
    numerical model (thresholds land exactly on training cuts)."""
    import torch

    import ydf_amd as ydf
    from ydf_amd.model.forest import (pack_binned8_nodes,
                                      padded_boundaries)

    assert torch.cuda.is_available()
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=40, validation_ratio=0).train(
        binary_data)
    dev = torch.device("cuda:0")
    X = torch.from_numpy(m._encode_features(binary_data)).to(dev)
    want = torch.from_numpy(m.predict(binary_data)).to(dev)
    bnd = padded_boundaries(m.dataspec.feature_columns)
    packed8 = torch.from_numpy(
        pack_binned8_nodes(m.forest, bnd,
                           leaf_scale=m._leaf_scale())).to(dev)
    bins = torch.empty(X.shape, dtype=torch.uint8, device=dev)
    ops.bin_data(X, torch.from_numpy(bnd).to(dev), bins)
    out = torch.empty(X.shape[1], dtype=torch.float32, device=dev)
    roots = torch.from_numpy(m.forest.roots).to(dev)
    ops.predict_forest_binned8(bins, packed8, roots, out,
                               init=float(m.init_predictions[0]))
    got = torch.sigmoid(out)
    assert (got - want).abs().max().item() < 1e-5


@pytest.mark.gpu
def test_binned8_engine_multiclass():
    """Compact-node engine with class-tree striding == default path."""
    import torch

    import ydf_amd as ydf

    assert torch.cuda.is_available()
    rng = np.random.RandomState(6)
    n = 70000  # above the auto-select batch threshold
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = np.where(x1 > 0.5, "a", np.where(x2 > 0, "b", "c"))
    d = {"x1": x1, "x2": x2, "label": y}
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20,
                                        validation_ratio=0).train(d)
    p_auto = m.predict(d)           # auto-selects binned8 (large batch)
    m.force_engine("flat")
    p_flat = m.predict(d)
    assert p_auto.shape == (n, 3)
    np.testing.assert_allclose(p_auto, p_flat, rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_weighted_poisson_histogram_integrity():
    """Advisor follow-up: weighted Poisson can reach per-example h*w
    near the packed-u64 limit; the post-weight clamp must keep the
    44-bit fixed-point h field sane. GPU and CPU models must agree in
    quality (corruption would destroy one of them)."""
    import torch

    import ydf_amd as ydf

    assert torch.cuda.is_available()
    rng = np.random.RandomState(8)
    n = 30000
    x = rng.randn(n).astype(np.float32)
    lam = np.exp(1.2 * x)
    w = rng.uniform(0.5, 50.0, n).astype(np.float32)  # rescaled to max 8
    d = {"x": x, "z": rng.randn(n).astype(np.float32),
         "w": w, "label": rng.poisson(lam).astype(np.float32)}
    kw = dict(label="label", task=ydf.Task.REGRESSION, loss="POISSON",
              weights="w", num_trees=25, validation_ratio=0.0)
    mg = ydf.GradientBoostedTreesLearner(**kw).train(d)
    mc = ydf.GradientBoostedTreesLearner(device="cpu", **kw).train(d)
    pg = mg.predict(d)
    pc = mc.predict(d)
    assert np.isfinite(pg).all() and (pg > 0).all()
    # both runs clamp identically; quality must match closely
    err_g = float(np.mean((pg - lam) ** 2))
    err_c = float(np.mean((pc - lam) ** 2))
    assert err_g < err_c * 1.3 + 1e-6, (err_g, err_c)


@pytest.mark.gpu
def test_binned4_engine_matches_flat(binary_data):
    """4-byte-node engine == flat engine on a trained model."""
    import torch

    import ydf_amd as ydf
    from ydf_amd.model.forest import (pack_binned4_nodes,
                                      padded_boundaries)

    assert torch.cuda.is_available()
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=40, validation_ratio=0).train(
        binary_data)
    dev = torch.device("cuda:0")
    X = torch.from_numpy(m._encode_features(binary_data)).to(dev)
    want = torch.from_numpy(m.predict(binary_data)).to(dev)
    bnd = padded_boundaries(m.dataspec.feature_columns)
    n4, lv4 = pack_binned4_nodes(m.forest, bnd,
                                 leaf_scale=m._leaf_scale())
    bins = torch.empty(X.shape, dtype=torch.uint8, device=dev)
    ops.bin_data(X, torch.from_numpy(bnd).to(dev), bins)
    out = torch.empty(X.shape[1], dtype=torch.float32, device=dev)
    ops.predict_forest_binned4(
        bins, torch.from_numpy(n4).to(dev),
        torch.from_numpy(lv4).to(dev),
        torch.from_numpy(m.forest.roots).to(dev), out,
        init=float(m.init_predictions[0]))
    got = torch.sigmoid(out)
    assert (got - want).abs().max().item() < 1e-5


@pytest.mark.gpu
def test_deep_rf_extraction_integrity():
    """Regression: pack_extract_kernel must cover node buffers larger
    than 64k slots (RF depth-16 trees) — a capped grid without a
    stride loop left the tail of the staging buffer stale, producing
    garbage feature indices (caught by tools/bench_rf.py)."""
    import ydf_amd as ydf

    rng = np.random.RandomState(0)
    n = 300000
    d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(10)}
    d["label"] = np.where(
        d["x0"] + d["x1"] * d["x2"] + 0.5 * rng.randn(n) > 0, "a", "b")
    m = ydf.RandomForestLearner(
        label="label", num_trees=3, max_depth=16,
        compute_oob_performances=False, device="cuda:0").train(d)
    f = m.forest
    valid = f.feat[f.feat >= 0]
    assert valid.size > 1000
    assert valid.max() < 10, int(valid.max())
    assert m.evaluate(d).accuracy > 0.85


@pytest.mark.gpu
def test_serving_session_graph_latency():
    """Graph-captured fixed-batch serving: bit-equal to predict() and
    substantially faster per small batch than the eager path."""
    import time

    import ydf_amd as ydf

    rng = np.random.RandomState(0)
    n = 50000
    d = {"x1": rng.randn(n).astype(np.float32),
         "x2": rng.randn(n).astype(np.float32),
         "x3": rng.randn(n).astype(np.float32)}
    d["label"] = np.where(d["x1"] - d["x2"] * d["x3"] > 0, "a", "b")
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=200, validation_ratio=0.0,
        device="cuda:0").train(d)
    B = 512
    batch = {k: v[:B] for k, v in d.items() if k != "label"}
    sess = m.serving_session(B)
    p_graph = sess.predict(batch)
    p_eager = m.predict(batch, device="cuda:0")
    np.testing.assert_allclose(p_graph, p_eager, rtol=1e-6, atol=1e-7)

    # timing, preformed-matrix form on both sides: the session replays
    # ONE graph; the eager path launches bin+walk(+activation) per call
    import torch

    X_np = m._encode_features(batch)
    np.testing.assert_allclose(sess.predict(X_np), p_eager, rtol=1e-6,
                               atol=1e-7)

    def eager_once():
        Xt = torch.from_numpy(X_np).to("cuda:0")
        out = m._apply_activation(m.predict_margin(Xt))
        return out.cpu().numpy()

    for _ in range(5):
        sess.predict(X_np)
        eager_once()
    t0 = time.perf_counter()
    for _ in range(50):
        sess.predict(X_np)
    t_graph = (time.perf_counter() - t0) / 50
    t0 = time.perf_counter()
    for _ in range(50):
        eager_once()
    t_eager = (time.perf_counter() - t0) / 50
    print(f"# serving latency: graph {t_graph*1e6:.0f}us vs eager "
          f"{t_eager*1e6:.0f}us per {B}-row batch")
    assert t_graph < t_eager, (t_graph, t_eager)


@pytest.mark.gpu
def test_small_batch_engine_equality():
    """The tree-parallel small-batch binned4 grid must match the flat
    engine exactly across tiny batch sizes (deterministic fixed-order
    partial reduction)."""
    import ydf_amd as ydf

    rng = np.random.RandomState(1)
    n = 100000
    d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(8)}
    d["label"] = np.where(d["x0"] - d["x1"] * d["x2"] > 0, "a", "b")
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=500, max_depth=6,
        validation_ratio=0.0, device="cuda:0").train(d)
    for B in (1, 7, 100, 1000, 30000):
        batch = {k: v[:B] for k, v in d.items() if k != "label"}
        p_auto = m.predict(batch, device="cuda:0")  # binned4(+tp)
        m.force_engine("flat")
        p_flat = m.predict(batch, device="cuda:0")
        m.force_engine(None)
        # chunked partial sums reduce in a different (fixed) order
        # than the flat walk -> ulp-level float differences only
        np.testing.assert_allclose(p_auto, p_flat, rtol=1e-5,
                                   atol=1e-6)
        # determinism across repeated calls is EXACT
        np.testing.assert_array_equal(
            p_auto, m.predict(batch, device="cuda:0"))


@pytest.mark.gpu
def test_small_batch_binned8_wide_model():
    """binned8 tree-parallel path for wide-feature models (>62
    features exceed the 4-byte packing): small batches must match the
    flat engine and stay deterministic."""
    import ydf_amd as ydf

    rng = np.random.RandomState(2)
    n = 60000
    F = 80
    d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(F)}
    d["label"] = np.where(
        d["x0"] + d["x1"] - d["x70"] + 0.3 * rng.randn(n) > 0, "a", "b")
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=300, max_depth=5,
        validation_ratio=0.0, device="cuda:0").train(d)
    from ydf_amd.model.forest import pack_binned4_nodes, padded_boundaries
    import pytest as _pt

    with _pt.raises(ValueError):
        pack_binned4_nodes(m.forest,
                           padded_boundaries(m.dataspec.feature_columns))
    for B in (100, 2000):
        batch = {k: v[:B] for k, v in d.items() if k != "label"}
        p_auto = m.predict(batch, device="cuda:0")
        m.force_engine("flat")
        p_flat = m.predict(batch, device="cuda:0")
        m.force_engine(None)
        np.testing.assert_allclose(p_auto, p_flat, rtol=1e-5, atol=1e-6)
        np.testing.assert_array_equal(
            p_auto, m.predict(batch, device="cuda:0"))


@pytest.mark.gpu
def test_small_batch_flat_engine_with_masks():
    """Flat-engine tree-parallel path for models with categorical
    masks (binned engines ineligible): small batches must match the
    large-batch flat walk exactly in structure (same kernel family,
    chunked reduction ulp tolerance) and improve latency."""
    import time

    import ydf_amd as ydf

    rng = np.random.RandomState(3)
    n = 80000
    d = {"x1": rng.randn(n).astype(np.float32),
         "x2": rng.randn(n).astype(np.float32),
         "c": rng.choice(["a", "b", "c", "d", "e", "f"], n)}
    d["label"] = np.where(
        (d["x1"] > 0) ^ np.isin(d["c"], ["a", "c"]), "p", "n")
    m = ydf.RandomForestLearner(
        label="label", num_trees=300, max_depth=10,
        compute_oob_performances=False, device="cuda:0").train(d)
    assert (m.forest.cat_idx >= 0).any()  # masks present -> flat engine
    big = {k: v[:40000] for k, v in d.items() if k != "label"}
    p_big = m.predict(big, device="cuda:0")
    for B in (100, 1000):
        batch = {k: v[:B] for k, v in d.items() if k != "label"}
        p_small = m.predict(batch, device="cuda:0")
        np.testing.assert_allclose(p_small, p_big[:B], rtol=1e-5,
                                   atol=1e-6)
        np.testing.assert_array_equal(
            p_small, m.predict(batch, device="cuda:0"))
    # latency sanity: 100-row batch should be far under the serial
    # whole-forest walk (~1 ms at 300 trees x depth 10)
    batch = {k: v[:100] for k, v in d.items() if k != "label"}
    for _ in range(3):
        m.predict(batch, device="cuda:0")
    t0 = time.perf_counter()
    for _ in range(20):
        m.predict(batch, device="cuda:0")
    t = (time.perf_counter() - t0) / 20
    print(f"# masked-model 100-row predict: {t*1e6:.0f}us")
    assert t < 0.002
