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
