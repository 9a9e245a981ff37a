"""Unit tests of the C++ CPU ops against plain numpy/torch references.

These same semantics are asserted against the HIP kernels on a GPU box in
tests/test_gpu_kernels.py (reference test analogue:
learner/decision_tree/decision_tree_test.cc brute-force splitter checks).
"""
import numpy as np
import torch

from ydf_amd import ops


def _rand_problem(seed=0, N=3000, F=4, n_bins=256):
    rng = np.random.RandomState(seed)
    bins = rng.randint(0, n_bins, size=(F, N)).astype(np.uint8)
    g = rng.randn(N).astype(np.float32)
    h = rng.rand(N).astype(np.float32) + 0.1
    node_ids = rng.randint(0, 4, size=N).astype(np.int32) + 3  # level 2
    return bins, g, h, node_ids


def test_grad_hess_binomial_matches_torch():
    rng = np.random.RandomState(0)
    preds = torch.from_numpy(rng.randn(1000).astype(np.float32))
    labels = torch.from_numpy(
        (rng.rand(1000) > 0.5).astype(np.float32))
    gh = torch.empty((1000, 2), dtype=torch.float32)
    ops.grad_hess(preds, labels, gh, 1)
    p = torch.sigmoid(preds)
    torch.testing.assert_close(gh[:, 0], p - labels, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(gh[:, 1], (p * (1 - p)).clamp_min(1e-16),
                               rtol=1e-5, atol=1e-6)


def test_grad_hess_softmax_matches_torch():
    rng = np.random.RandomState(0)
    C, N = 4, 500
    preds = torch.from_numpy(rng.randn(C, N).astype(np.float32))
    labels = torch.from_numpy(rng.randint(0, C, N).astype(np.float32))
    gh = torch.empty((N, 2), dtype=torch.float32)
    ops.grad_hess_softmax(preds, labels, gh, C, 2)
    p = torch.softmax(preds, dim=0)[2]
    y = (labels == 2).float()
    torch.testing.assert_close(gh[:, 0], p - y, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(gh[:, 1], (p * (1 - p)).clamp_min(1e-16),
                               rtol=1e-4, atol=1e-5)


def test_hist_build_matches_numpy():
    bins, g, h, node_ids = _rand_problem()
    F, N = bins.shape
    n_bins = 256
    n_slots = 4
    gh = torch.from_numpy(np.stack([g, h], axis=1).copy())
    hist = torch.zeros((n_slots, F, n_bins, 3), dtype=torch.float32)
    slot_map = torch.arange(4, dtype=torch.int32)
    ops.hist_build(torch.from_numpy(bins), gh,
                   torch.from_numpy(node_ids), slot_map, hist,
                   level_base=3, level_size=4, slot0=0, n_slots=4)
    # numpy reference
    ref = np.zeros((n_slots, F, n_bins, 3), dtype=np.float64)
    for i in range(N):
        s = node_ids[i] - 3
        for f in range(F):
            ref[s, f, bins[f, i], 0] += g[i]
            ref[s, f, bins[f, i], 1] += h[i]
            ref[s, f, bins[f, i], 2] += 1
    np.testing.assert_allclose(hist.numpy(), ref, rtol=1e-4, atol=1e-4)


def test_hist_build_skips_zero_weight_counts():
    bins = np.zeros((1, 4), dtype=np.uint8)
    gh = torch.tensor([[1.0, 0.0], [1.0, 2.0], [0.5, 1.0], [0.25, 0.0]])
    node_ids = torch.zeros(4, dtype=torch.int32)
    hist = torch.zeros((1, 1, 256, 3), dtype=torch.float32)
    ops.hist_build(torch.from_numpy(bins), gh, node_ids,
                   torch.zeros(1, dtype=torch.int32), hist, 0, 1, 0, 1)
    assert hist[0, 0, 0, 2].item() == 2.0  # only h != 0 rows counted
    assert abs(hist[0, 0, 0, 1].item() - 3.0) < 1e-6


def _brute_force_best_split(hist_nf, lam, min_ex, min_h):
    """Reference scan: hist_nf [F, B, 3] -> (gain, feat, bin). Totals are
    per-feature (identical across features for any real histogram)."""
    F, B, _ = hist_nf.shape
    best = (-np.inf, -1, 0)
    for f in range(F):
        G, H, C = hist_nf[f].sum(axis=0)
        parent = G * G / (H + lam)
        cg = np.cumsum(hist_nf[f, :, 0])
        ch = np.cumsum(hist_nf[f, :, 1])
        cc = np.cumsum(hist_nf[f, :, 2])
        for b in range(B - 1):
            CL, CR = cc[b], C - cc[b]
            HL, HR = ch[b], H - ch[b]
            if CL < min_ex or CR < min_ex or HL < min_h or HR < min_h:
                continue
            gain = cg[b] ** 2 / (HL + lam) + (G - cg[b]) ** 2 / (HR + lam) \
                - parent
            if gain > best[0]:
                best = (gain, f, b)
    return best


def test_split_scan_matches_brute_force():
    rng = np.random.RandomState(3)
    F, B = 5, 256
    n_slots = 3
    hist = rng.rand(n_slots, F, B, 3).astype(np.float32)
    hist[..., 2] = rng.randint(0, 10, size=(n_slots, F, B))
    # make counts consistent across features per slot (same row totals):
    for s in range(n_slots):
        for f in range(1, F):
            hist[s, f, :, 2] = hist[s, 0, :, 2][
                rng.permutation(B)]
    ht = torch.from_numpy(hist)
    total_nodes = 31
    node_stats = torch.zeros((total_nodes, 3), dtype=torch.float32)
    bg = torch.empty((n_slots, F), dtype=torch.float32)
    bb = torch.empty((n_slots, F), dtype=torch.int32)
    bf = torch.empty(n_slots, dtype=torch.int32)
    bbin = torch.empty(n_slots, dtype=torch.int32)
    bgain = torch.empty(n_slots, dtype=torch.float32)
    abs_of_slot = torch.tensor([3, 4, 5], dtype=torch.int32)
    lam, min_ex, min_h = 1.0, 5, 0.0
    ops.split_scan(ht, abs_of_slot, node_stats, bg, bb, bf, bbin, bgain,
                   0, n_slots, lam, min_h, min_ex, 0.0)
    for s in range(n_slots):
        gain, f, b = _brute_force_best_split(hist[s], lam, min_ex, min_h)
        assert bf[s].item() == f
        assert bbin[s].item() == b
        np.testing.assert_allclose(bgain[s].item(), gain, rtol=1e-3)
        # child stats consistency
        abs_node = 3 + s
        tot = node_stats[abs_node].numpy()
        lc = node_stats[2 * abs_node + 1].numpy()
        rc = node_stats[2 * abs_node + 2].numpy()
        np.testing.assert_allclose(lc + rc, tot, rtol=1e-3, atol=1e-3)


def test_update_node_ids_and_parking():
    N = 100
    bins = np.zeros((1, N), dtype=np.uint8)
    bins[0, :50] = 10
    bins[0, 50:] = 200
    node_ids = torch.zeros(N, dtype=torch.int32)
    slot_map = torch.zeros(1, dtype=torch.int32)
    best_feat = torch.tensor([0], dtype=torch.int32)
    best_bin = torch.tensor([100], dtype=torch.int32)
    ops.update_node_ids(torch.from_numpy(bins), node_ids, slot_map,
                        best_feat, best_bin, 0, 1)
    assert (node_ids[:50] == 1).all()   # bin 10 <= 100 -> left
    assert (node_ids[50:] == 2).all()   # bin 200 > 100 -> right
    # leaf node parks
    best_feat2 = torch.tensor([-1, -1], dtype=torch.int32)
    best_bin2 = torch.tensor([0, 0], dtype=torch.int32)
    before = node_ids.clone()
    ops.update_node_ids(torch.from_numpy(bins), node_ids,
                        torch.arange(2, dtype=torch.int32), best_feat2,
                        best_bin2, 1, 2)
    assert torch.equal(node_ids, before)


def test_predict_forest_matches_python_walk():
    rng = np.random.RandomState(5)
    F, N = 3, 500
    X = rng.randn(F, N).astype(np.float32)
    # two hand-built trees
    #  tree0: root(x0 > 0.0) -> leaves 0.5 / -1.5
    #  tree1: root(x2 > 0.3) -> (x1 > -0.2 -> 2.0/0.25) / leaf 1.0
    feat = np.array([0, -1, -1, 2, -1, 1, -1, -1], dtype=np.int32)
    thr = np.array([0.0, -1.5, 0.5, 0.3, 1.0, -0.2, 0.25, 2.0],
                   dtype=np.float32)
    left = np.array([1, 0, 0, 4, 0, 6, 0, 0], dtype=np.int32)
    roots = np.array([0, 3], dtype=np.int32)
    out = torch.empty(N)
    ops.predict_forest(torch.from_numpy(X), torch.from_numpy(feat),
                       torch.from_numpy(thr), torch.from_numpy(left),
                       torch.from_numpy(roots), out, init=0.25, scale=1.0)

    def walk(i):
        acc = 0.25
        for r in roots:
            n = r
            while feat[n] >= 0:
                n = left[n] + (1 if X[feat[n], i] > thr[n] else 0)
            acc += thr[n]
        return acc

    ref = np.array([walk(i) for i in range(N)], dtype=np.float32)
    np.testing.assert_allclose(out.numpy(), ref, rtol=1e-6, atol=1e-6)


def test_weighted_target():
    y = torch.tensor([1.0, 0.0, 1.0])
    w = torch.tensor([2.0, 1.0, 0.0])
    gh = torch.empty((3, 2))
    ops.weighted_target(y, w, gh)
    np.testing.assert_allclose(gh.numpy(),
                               [[-2.0, 2.0], [0.0, 1.0], [0.0, 0.0]])
    ops.weighted_target(y, None, gh)
    np.testing.assert_allclose(gh.numpy(),
                               [[-1.0, 1.0], [0.0, 1.0], [-1.0, 1.0]])


def test_binary_logloss_matches_torch():
    rng = np.random.RandomState(0)
    preds = torch.from_numpy(rng.randn(2000).astype(np.float32) * 3)
    labels = torch.from_numpy((rng.rand(2000) > 0.4).astype(np.float32))
    out = torch.zeros(2)
    ops.binary_logloss(preds, labels, out)
    ref_loss = torch.nn.functional.binary_cross_entropy_with_logits(
        preds, labels, reduction="sum")
    ref_acc = (((preds > 0) == (labels > 0.5)).float().sum())
    np.testing.assert_allclose(out[0].item(), ref_loss.item(), rtol=1e-4)
    np.testing.assert_allclose(out[1].item(), ref_acc.item(), rtol=0)


def test_bin_data_na_flag():
    """bin_data(na_to_255): NaN rows land in the reserved bin, other
    values keep the standard cut semantics."""
    x = torch.tensor([[0.1, np.nan, 5.0, -3.0]], dtype=torch.float32)
    bnd = torch.tensor([[0.0, 1.0, 2.0]], dtype=torch.float32)
    out = torch.empty((1, 4), dtype=torch.uint8)
    ops.bin_data(x, bnd, out, na_to_255=True)
    assert out.tolist() == [[1, 255, 3, 0]]
    ops.bin_data(x, bnd, out)  # default: NaN compares false -> bin 0
    assert out[0, 1].item() == 0


def test_split_scan_l1_shrinks_gain():
    """lambda_l1 soft-thresholds gradient sums: a weak split's gain hits
    zero once l1 exceeds |G| on both sides."""
    n_bins = ops.MAX_BINS
    hist = torch.zeros((1, 1, n_bins, 3), dtype=torch.float32)
    hist[0, 0, 10] = torch.tensor([-1.0, 4.0, 8.0])
    hist[0, 0, 200] = torch.tensor([1.5, 4.0, 8.0])
    args = dict(
        abs_of_slot=torch.zeros(1, dtype=torch.int32),
        node_stats=torch.zeros((3, 3), dtype=torch.float32),
        best_gain_nf=torch.zeros((1, 1), dtype=torch.float32),
        best_bin_nf=torch.zeros((1, 1), dtype=torch.int32),
        best_feat=torch.zeros(1, dtype=torch.int32),
        best_bin=torch.zeros(1, dtype=torch.int32),
        best_gain=torch.zeros(1, dtype=torch.float32))
    ops.split_scan(hist, args["abs_of_slot"], args["node_stats"],
                   args["best_gain_nf"], args["best_bin_nf"],
                   args["best_feat"], args["best_bin"],
                   args["best_gain"], 0, 1, 0.0, 0.0, 1, 0.0)
    g0 = float(args["best_gain"][0])
    assert g0 > 0
    ops.split_scan(hist, args["abs_of_slot"], args["node_stats"],
                   args["best_gain_nf"], args["best_bin_nf"],
                   args["best_feat"], args["best_bin"],
                   args["best_gain"], 0, 1, 0.0, 0.0, 1, 0.0,
                   lambda_l1=0.5)
    g1 = float(args["best_gain"][0])
    assert 0 <= g1 < g0
    ops.split_scan(hist, args["abs_of_slot"], args["node_stats"],
                   args["best_gain_nf"], args["best_bin_nf"],
                   args["best_feat"], args["best_bin"],
                   args["best_gain"], 0, 1, 0.0, 0.0, 1, 0.0,
                   lambda_l1=10.0)
    assert int(args["best_feat"][0]) == -1  # fully suppressed
