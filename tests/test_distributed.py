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
