"""Cox proportional-hazards survival training (reference
loss_imp_cox.cc, task SURVIVAL_ANALYSIS) + concordance index."""
import numpy as np
import pytest

import ydf_amd as ydf
from ydf_amd.metric.survival import concordance_index


def _surv_data(n=8000, seed=0):
    rng = np.random.RandomState(seed)
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    hazard = np.exp(x1 - 0.5 * x2)
    T = rng.exponential(1.0 / hazard)
    C = rng.exponential(2.0, n)
    time = np.minimum(T, C)
    event = T <= C
    return ({"x1": x1, "x2": x2, "time": time.astype(np.float32),
             "event": event}, time, event, hazard, rng)


def test_cox_gbt_learns_hazard():
    d, time, event, hazard, rng = _surv_data()
    m = ydf.GradientBoostedTreesLearner(
        label="time", label_event_observed="event",
        task=ydf.Task.SURVIVAL_ANALYSIS, num_trees=50,
        validation_ratio=0.1).train(d)
    ev = m.evaluate(d)
    oracle = concordance_index(time, event, np.log(hazard))
    assert ev.cindex > oracle - 0.03
    assert ev.cindex > 0.7
    # higher x1 -> higher predicted log-hazard
    p = m.predict(d)
    assert np.corrcoef(p, np.log(hazard))[0, 1] > 0.8


def test_cox_persistence(tmp_path):
    d, *_ = _surv_data(2000, 1)
    m = ydf.GradientBoostedTreesLearner(
        label="time", label_event_observed="event",
        task=ydf.Task.SURVIVAL_ANALYSIS, num_trees=10,
        validation_ratio=0).train(d)
    p1 = m.predict(d)
    m.save(str(tmp_path / "cox"))
    m2 = ydf.load_model(str(tmp_path / "cox"))
    np.testing.assert_array_equal(p1, m2.predict(d))
    assert m2.task() == ydf.Task.SURVIVAL_ANALYSIS
    assert m2.evaluate(d).cindex > 0.6


def test_cox_left_truncation():
    """entry age (left truncation) restricts risk sets; training stays
    sane."""
    d, time, event, hazard, rng = _surv_data(4000, 2)
    entry = (time * rng.uniform(0, 0.5, len(time))).astype(np.float32)
    d["entry"] = entry
    m = ydf.GradientBoostedTreesLearner(
        label="time", label_event_observed="event",
        label_entry_age="entry",
        task=ydf.Task.SURVIVAL_ANALYSIS, num_trees=20,
        validation_ratio=0).train(d)
    assert np.corrcoef(m.predict(d), np.log(hazard))[0, 1] > 0.7


def test_concordance_index_properties():
    rng = np.random.RandomState(3)
    t = rng.exponential(1, 500)
    e = rng.rand(500) < 0.7
    assert abs(concordance_index(t, e, rng.randn(500)) - 0.5) < 0.06
    # perfect anti-time score (hazard = -time) is concordant
    assert concordance_index(t, e, -t) == 1.0
    assert concordance_index(t, e, t) == 0.0
