"""Large-vocabulary categorical training (vocab > 256).

Reference analogue: full-dictionary CART ordering per node
(splitter_scanner.h:859) with max_vocab_count up to 2000
(data_spec.proto:49-61). The MI355X path trains on <=256 CART-ordered
category GROUPS (mass-balanced over the global mean-label ordering) and
expands chosen group masks back to full-dictionary set conditions; the
pre-round-2 behavior collapsed every code >= 255 into one shared bin.
"""
import numpy as np
import pytest

import ydf_amd as ydf


def _bigcat_data(n=40000, V=2000, seed=0):
    rng = np.random.RandomState(seed)
    cats = rng.randint(0, V, n)
    # label depends on the individual category (odd/even + noise), so
    # any two categories collapsed together lose signal
    p = np.where(cats % 2 == 0, 0.9, 0.1)
    y = rng.rand(n) < p
    names = np.array([f"c{i:04d}" for i in range(V)])
    return {"cat": names[cats], "x": rng.randn(n).astype(np.float32),
            "label": np.where(y, "pos", "neg")}


def test_bigcat_gbt_quality_and_serving():
    data = _bigcat_data()
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, max_depth=4, validation_ratio=0.0,
        device="cpu").train(data)
    acc = m.evaluate(data).accuracy
    # the old >=255-shared-bin collapse caps accuracy near the base
    # rate for the ~87% of rows whose category exceeds code 255
    assert acc > 0.85, acc
    # serving path must agree with itself after save/load
    p = m.predict(data)
    assert m.forest.has_set_conditions  # full-dictionary conditions


def test_bigcat_save_load_roundtrip(tmp_path):
    data = _bigcat_data(n=8000, V=600, seed=1)
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=8, max_depth=4, validation_ratio=0.0,
        device="cpu").train(data)
    p1 = m.predict(data)
    out = str(tmp_path / "m")
    m.save(out)
    m2 = ydf.load_model(out)
    p2 = m2.predict(data)
    np.testing.assert_allclose(p1, p2, rtol=1e-5, atol=1e-6)


def test_bigcat_rf():
    data = _bigcat_data(n=20000, V=500, seed=2)
    m = ydf.RandomForestLearner(
        label="label", num_trees=10, max_depth=8,
        device="cpu").train(data)
    assert m.evaluate(data).accuracy > 0.85


def test_small_vocab_unchanged():
    """Vocab <= 256 must keep the direct code==bin path (no grouping,
    no set conditions)."""
    rng = np.random.RandomState(3)
    n = 5000
    cats = rng.randint(0, 20, n)
    data = {"cat": np.array([f"c{c}" for c in cats]),
            "label": np.where(cats % 2 == 0, "a", "b")}
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=5, validation_ratio=0.0,
        device="cpu").train(data)
    assert not m.forest.has_set_conditions
    assert m.evaluate(data).accuracy > 0.99
