"""Data-parallel training tests on CPU with the gloo backend, world_size=2.

The same code path runs RCCL over xGMI on an 8-GPU MI355X node (one process
per GPU); these tests pin the collective semantics without a GPU
(reference analogue: the MULTI_THREAD distribute implementation used by
distributed_gradient_boosted_trees_test.cc:59-134).
"""
import multiprocessing as mp
import os
import pickle
import socket

import numpy as np
import pytest

import ydf_amd as ydf


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _make_data(n=6000):
    rng = np.random.RandomState(0)
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    x3 = rng.randn(n).astype(np.float32)
    y = (2 * x1 - x2 + 0.5 * x1 * x2 > 0)
    return {"x1": x1, "x2": x2, "x3": x3,
            "label": np.where(y, "yes", "no")}


def _rf_args():
    return dict(label="label", num_trees=4, max_depth=6,
                bootstrap_training_dataset=False,
                num_candidate_attributes=-1, min_examples=5, device="cpu")


def _worker_rf(rank, world, port, out_path):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ydf_amd.parallel.dist import shard_rows

        data = _make_data()
        ds = ydf.create_vertical_dataset(data, label="label",
                                         task=ydf.Task.CLASSIFICATION)
        lo, hi = shard_rows(ds.n_examples, rank, world)
        m = ydf.RandomForestLearner(**_rf_args()).train(ds.shard(lo, hi))
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump({"feat": m.forest.feat, "thr": m.forest.thr,
                             "left": m.forest.left,
                             "roots": m.forest.roots}, f)
    finally:
        dist.destroy_process_group()


def _worker_gbt(rank, world, port, out_path):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ydf_amd.parallel.dist import shard_rows

        data = _make_data()
        ds = ydf.create_vertical_dataset(data, label="label",
                                         task=ydf.Task.CLASSIFICATION)
        lo, hi = shard_rows(ds.n_examples, rank, world)
        m = ydf.GradientBoostedTreesLearner(
            label="label", num_trees=30, validation_ratio=0.1,
            device="cpu").train(ds.shard(lo, hi))
        if rank == 0:
            preds = m.predict(data, device="cpu")
            # model predicts P(label_classes[1]); vocab is frequency-ordered
            labels = (np.asarray(data["label"]) == m.label_classes[1])
            from ydf_amd.metric.metric import roc_auc

            with open(out_path, "wb") as f:
                pickle.dump({"auc": roc_auc(labels, preds),
                             "n_trees": m.num_trees()}, f)
    finally:
        dist.destroy_process_group()


def _spawn(target, out_path, world=2):
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=target, args=(r, world, port, out_path))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def test_rf_distributed_bit_exact(tmp_path):
    """Integer-valued gradient sums make shard all-reduce exact, so the
    2-rank forest must be IDENTICAL to single-process training."""
    out = str(tmp_path / "rf.pkl")
    _spawn(_worker_rf, out)
    with open(out, "rb") as f:
        got = pickle.load(f)
    data = _make_data()
    m = ydf.RandomForestLearner(**_rf_args()).train(data)
    np.testing.assert_array_equal(got["feat"], m.forest.feat)
    np.testing.assert_array_equal(got["left"], m.forest.left)
    np.testing.assert_array_equal(got["roots"], m.forest.roots)
    np.testing.assert_allclose(got["thr"], m.forest.thr, rtol=1e-6)


def test_rf_distributed_dense_compacted_allreduce(tmp_path):
    """Dense levels under histogram subtraction all-reduce only the
    BUILT slots (derived siblings reduce locally): the 2-rank forest
    must still equal single-process training bit for bit."""
    out = str(tmp_path / "rfd.pkl")
    _spawn(_worker_rf_dense, out)
    with open(out, "rb") as f:
        got = pickle.load(f)
    os.environ["YDFA_DENSE_LIMIT"] = "8"
    try:
        m = ydf.RandomForestLearner(**_rf_args()).train(_make_data())
    finally:
        os.environ.pop("YDFA_DENSE_LIMIT", None)
    np.testing.assert_array_equal(got["feat"], m.forest.feat)
    np.testing.assert_array_equal(got["left"], m.forest.left)
    np.testing.assert_allclose(got["thr"], m.forest.thr, rtol=1e-6)


def test_gbt_distributed_quality(tmp_path):
    out = str(tmp_path / "gbt.pkl")
    _spawn(_worker_gbt, out)
    with open(out, "rb") as f:
        got = pickle.load(f)
    assert got["auc"] > 0.97, got
    assert got["n_trees"] >= 10


def _worker_rf_dense(rank, world, port, out_path):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["YDFA_DENSE_LIMIT"] = "8"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ydf_amd.parallel.dist import shard_rows

        data = _make_data()
        ds = ydf.create_vertical_dataset(data, label="label",
                                         task=ydf.Task.CLASSIFICATION)
        lo, hi = shard_rows(ds.n_examples, rank, world)
        m = ydf.RandomForestLearner(**_rf_args()).train(ds.shard(lo, hi))
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump({"feat": m.forest.feat, "thr": m.forest.thr,
                             "left": m.forest.left}, f)
    finally:
        os.environ.pop("YDFA_DENSE_LIMIT", None)
        dist.destroy_process_group()


def _worker_oblique(rank, world, port, out_path):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ydf_amd.parallel.dist import shard_rows

        data = _make_data()
        ds = ydf.create_vertical_dataset(data, label="label",
                                         task=ydf.Task.CLASSIFICATION)
        lo, hi = shard_rows(ds.n_examples, rank, world)
        m = ydf.GradientBoostedTreesLearner(
            label="label", num_trees=8, max_depth=4, validation_ratio=0,
            split_axis="SPARSE_OBLIQUE",
            device="cpu").train(ds.shard(lo, hi))
        # EVERY rank writes its forest: ranks must agree bit-for-bit
        with open(f"{out_path}.{rank}", "wb") as f:
            pickle.dump({"feat": m.forest.feat, "thr": m.forest.thr,
                         "obl_w": m.forest.obl_w,
                         "obl_attr": m.forest.obl_attr}, f)
    finally:
        dist.destroy_process_group()


def test_oblique_distributed_rank_consistent(tmp_path):
    """Oblique projections are seeded per (tree, level) and bin cuts are
    broadcast from rank 0, so all ranks must build the same model."""
    out = str(tmp_path / "obl.pkl")
    _spawn(_worker_oblique, out)
    with open(out + ".0", "rb") as f:
        a = pickle.load(f)
    with open(out + ".1", "rb") as f:
        b = pickle.load(f)
    np.testing.assert_array_equal(a["feat"], b["feat"])
    np.testing.assert_array_equal(a["thr"], b["thr"])
    np.testing.assert_array_equal(a["obl_attr"], b["obl_attr"])
    np.testing.assert_array_equal(a["obl_w"], b["obl_w"])
    assert len(a["obl_attr"]) > 0


# ---------------------------------------------------------------------------
# Fault injection + recovery (reference simulate_worker_failure,
# distributed_gradient_boosted_trees_test.cc:148-181 + worker.h:121
# MaybeSimulateFailure: a worker dies mid-training, the job restarts and
# must resume from the last checkpoint and converge)
# ---------------------------------------------------------------------------
def _worker_gbt_snap(rank, world, port, out_path, workdir, fault_iter):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    if fault_iter >= 0:
        os.environ["YDFA_FAULT_ITER"] = str(fault_iter)
        os.environ["YDFA_FAULT_RANK"] = "1"
    else:
        os.environ.pop("YDFA_FAULT_ITER", None)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from ydf_amd.parallel.dist import shard_rows

        data = _make_data()
        ds = ydf.create_vertical_dataset(data, label="label",
                                         task=ydf.Task.CLASSIFICATION)
        lo, hi = shard_rows(ds.n_examples, rank, world)
        m = ydf.GradientBoostedTreesLearner(
            label="label", num_trees=24, validation_ratio=0.0,
            early_stopping="NONE",
            working_dir=workdir, resume_training=True,
            resume_training_snapshot_interval_seconds=0.0,
            device="cpu").train(ds.shard(lo, hi))
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump({"preds": m.predict(data, device="cpu"),
                             "n_trees": m.num_trees()}, f)
    finally:
        dist.destroy_process_group()


def _spawn_may_fail(target, args, world=2):
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=target, args=(r, world, port) + args)
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    return [p.exitcode for p in procs]


def test_distributed_fault_recovery(tmp_path):
    """Rank 1 dies at iteration 8 (snapshots every iteration); the
    restarted job must resume from the snapshot and produce the same
    forest as an uninterrupted run."""
    wd_f = str(tmp_path / "wd_fault")
    out = str(tmp_path / "snap.pkl")

    codes = _spawn_may_fail(_worker_gbt_snap,
                            (out, wd_f, 8))
    assert any(c != 0 for c in codes), "injected fault did not fire"
    snap = os.path.join(wd_f, "snapshot")
    assert os.path.exists(os.path.join(snap, "done")), \
        "no snapshot written before the fault"

    # restart (no fault): resumes from snapshot
    codes = _spawn_may_fail(_worker_gbt_snap, (out, wd_f, -1))
    assert codes == [0, 0], codes
    with open(out, "rb") as f:
        resumed = pickle.load(f)
    assert resumed["n_trees"] == 24

    # uninterrupted reference run in a fresh working dir
    wd_c = str(tmp_path / "wd_clean")
    out2 = str(tmp_path / "clean.pkl")
    codes = _spawn_may_fail(_worker_gbt_snap, (out2, wd_c, -1))
    assert codes == [0, 0], codes
    with open(out2, "rb") as f:
        clean = pickle.load(f)
    assert clean["n_trees"] == 24
    # margins are re-summed on resume -> ulp differences may flip
    # near-tie splits for a handful of examples (same tolerance as the
    # single-process exact-resume test)
    p2, p3 = resumed["preds"], clean["preds"]
    assert np.mean(np.abs(p2 - p3) < 1e-4) > 0.99
    assert np.abs(p2 - p3).max() < 0.2


# ---------------------------------------------------------------------------
# torchrun rehearsals: the exact launch path the driver uses for the
# 8-GPU scaling bench, exercised end-to-end (world-size-1 initializes a
# REAL process group; on a GPU box that is a real RCCL communicator +
# ncclAllReduce calls)
# ---------------------------------------------------------------------------
def _run_torchrun_bench(nproc, extra_env=None):
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update(extra_env or {})
    cmd = [sys.executable, "-m", "torch.distributed.run", "--standalone",
           "--local-addr", "127.0.0.1", f"--nproc-per-node={nproc}",
           os.path.join(repo, "bench.py"), "--gpus", str(nproc),
           "--rows", "200000", "--steps", "4", "--warmup", "1"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=repo, env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    import json as _json

    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith("{")][-1]
    return _json.loads(line)


def test_torchrun_world1_cpu():
    out = _run_torchrun_bench(1, {"YDFA_DIST_BACKEND": "gloo"})
    assert out["n_gpus"] == 1
    assert out["value"] > 0


@pytest.mark.gpu
def test_torchrun_world1_rccl():
    """Single-rank torchrun on a real GPU: full RCCL init + world-size-1
    ncclAllReduce path — the rehearsal that de-risks the driver's 8-GPU
    launch (VERDICT round-1 item #1)."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    out = _run_torchrun_bench(1)
    assert out["n_gpus"] == 1
    assert out["value"] > 0


@pytest.mark.gpu
def test_torchrun_oversubscribed_gloo_fallback():
    """2 ranks on 1 GPU: RCCL would refuse (Duplicate GPU); the gloo
    fallback must keep the rehearsal runnable end-to-end."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    if torch.cuda.device_count() >= 2:
        pytest.skip("only meaningful when ranks exceed GPUs")
    out = _run_torchrun_bench(2)
    assert out["n_gpus"] == 2
    assert out["value"] > 0
