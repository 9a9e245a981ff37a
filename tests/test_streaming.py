"""Out-of-core (>HBM) training over the on-disk binned cache
(reference ShardedSamplingTrain, gradient_boosted_trees.cc:655 +
dataset_cache.h:15-58)."""
import numpy as np
import pytest
import torch

import ydf_amd as ydf


def _data(n=30000, seed=0):
    rng = np.random.RandomState(seed)
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    cat = rng.choice(["a", "b", "c", "d"], n)
    y = (2 * x1 - x2 + np.where(cat == "a", 1.5, 0.0)
         + 0.3 * rng.randn(n)) > 0
    return {"x1": x1, "x2": x2, "cat": cat,
            "label": np.where(y, "p", "n")}


def test_streaming_gbt_matches_in_memory(tmp_path):
    """Chunked out-of-core training (7 chunks) must match in-memory
    training up to cross-chunk float accumulation order."""
    data = _data()
    cache = ydf.create_dataset_cache(data, str(tmp_path / "cache"),
                                     label="label", chunk_rows=4500)
    assert cache.n_chunks == 7
    kw = dict(label="label", num_trees=20, max_depth=4,
              validation_ratio=0.0, device="cpu")
    m_str = ydf.GradientBoostedTreesLearner(**kw).train(cache)
    m_mem = ydf.GradientBoostedTreesLearner(**kw).train(data)
    acc_s = m_str.evaluate(data).accuracy
    acc_m = m_mem.evaluate(data).accuracy
    assert abs(acc_s - acc_m) < 0.01, (acc_s, acc_m)
    assert acc_s > 0.9
    # chunk-order float accumulation can flip near-tie splits; the
    # prediction distribution must still agree closely
    p_s = m_str.predict(data)
    p_m = m_mem.predict(data)
    assert np.mean(np.abs(p_s - p_m) < 1e-3) > 0.98


def test_streaming_regression(tmp_path):
    rng = np.random.RandomState(1)
    n = 20000
    x = rng.randn(n).astype(np.float32)
    d = {"x": x, "z": rng.randn(n).astype(np.float32),
         "label": (3 * x + 0.1 * rng.randn(n)).astype(np.float32)}
    cache = ydf.create_dataset_cache(d, str(tmp_path / "c"),
                                     label="label",
                                     task=ydf.Task.REGRESSION,
                                     chunk_rows=6000)
    m = ydf.GradientBoostedTreesLearner(
        label="label", task=ydf.Task.REGRESSION, num_trees=30,
        validation_ratio=0.0, device="cpu").train(cache)
    assert m.evaluate(d).rmse < 0.6


def test_cache_single_chunk_identical(tmp_path):
    """One chunk: the streamed model must pick the SAME splits as the
    in-memory model (identical structure); leaf values agree up to the
    fp accumulation differences between the dense in-memory kernels
    (histogram subtraction) and the always-direct streamed build."""
    data = _data(n=8000, seed=2)
    cache = ydf.create_dataset_cache(data, str(tmp_path / "c1"),
                                     label="label", chunk_rows=10**6)
    kw = dict(label="label", num_trees=10, max_depth=4,
              validation_ratio=0.0, device="cpu")
    m_str = ydf.GradientBoostedTreesLearner(**kw).train(cache)
    m_mem = ydf.GradientBoostedTreesLearner(**kw).train(data)
    np.testing.assert_array_equal(m_str.forest.feat, m_mem.forest.feat)
    np.testing.assert_array_equal(m_str.forest.left, m_mem.forest.left)
    np.testing.assert_allclose(m_str.forest.thr, m_mem.forest.thr,
                               rtol=2e-3, atol=2e-3)
    np.testing.assert_allclose(m_str.predict(data), m_mem.predict(data),
                               atol=5e-3)


@pytest.mark.gpu
def test_streaming_gbt_gpu(tmp_path):
    assert torch.cuda.is_available()
    data = _data(n=50000, seed=3)
    cache = ydf.create_dataset_cache(data, str(tmp_path / "cg"),
                                     label="label", chunk_rows=8000)
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, max_depth=4,
        validation_ratio=0.0, device="cuda:0").train(cache)
    assert m.evaluate(data).accuracy > 0.9


def _worker_streaming(rank, world, port, cache_dir, out_path):
    import os

    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import pickle

        import ydf_amd as ydf
        from ydf_amd.dataset.cache import DatasetCache

        cache = DatasetCache(cache_dir)
        m = ydf.GradientBoostedTreesLearner(
            label="label", num_trees=10, max_depth=4,
            validation_ratio=0.0, device="cpu").train(cache)
        with open(f"{out_path}.{rank}", "wb") as f:
            pickle.dump({"feat": m.forest.feat, "thr": m.forest.thr},
                        f)
    finally:
        dist.destroy_process_group()


def test_streaming_distributed_chunk_sharded(tmp_path):
    """Chunk-sharded data-parallel streaming: 2 ranks each own half
    the chunks; per-level histograms all-reduce, so both ranks must
    build the same forest, and it must match the single-process
    streamed forest (same reduced histograms up to fp order)."""
    import multiprocessing as mp
    import pickle
    import socket

    data = _data(n=24000, seed=7)
    cache = ydf.create_dataset_cache(data, str(tmp_path / "cache"),
                                     label="label", chunk_rows=4000)
    assert cache.n_chunks == 6

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    out = str(tmp_path / "m")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_streaming,
                         args=(r, 2, port, cache.path, out))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    with open(out + ".0", "rb") as f:
        a = pickle.load(f)
    with open(out + ".1", "rb") as f:
        b = pickle.load(f)
    np.testing.assert_array_equal(a["feat"], b["feat"])
    np.testing.assert_allclose(a["thr"], b["thr"], rtol=1e-6)
    # single-process reference
    m1 = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=10, max_depth=4, validation_ratio=0.0,
        device="cpu").train(cache)
    np.testing.assert_array_equal(a["feat"], m1.forest.feat)
    np.testing.assert_allclose(a["thr"], m1.forest.thr, rtol=5e-3,
                               atol=5e-3)
