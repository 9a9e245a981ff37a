"""Model persistence + inspection tests (reference analogue:
model serialization round-trip in PostTrainingChecks, utils/test_utils.h)."""
import os

import numpy as np
import pytest

import ydf_amd as ydf
from ydf_amd.model import tree as tree_lib


@pytest.fixture(scope="module")
def trained(binary_data):
    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=20).train(
        binary_data)
    return m


def test_save_load_roundtrip(tmp_path, trained, binary_data):
    p = str(tmp_path / "model")
    trained.save(p)
    assert os.path.exists(os.path.join(p, "done"))
    m2 = ydf.load_model(p)
    assert isinstance(m2, ydf.GradientBoostedTreesModel)
    np.testing.assert_allclose(trained.predict(binary_data),
                               m2.predict(binary_data), rtol=1e-6, atol=1e-7)
    assert m2.label() == "income" or m2.label() == "label"
    assert m2.task() == ydf.Task.CLASSIFICATION
    assert m2.num_trees() == trained.num_trees()


def test_serialize_roundtrip(trained, binary_data):
    blob = ydf.serialize_model(trained)
    m2 = ydf.deserialize_model(blob)
    np.testing.assert_allclose(trained.predict(binary_data),
                               m2.predict(binary_data), rtol=1e-6, atol=1e-7)


def test_tree_inspection(trained):
    t = trained.get_tree(0)
    assert isinstance(t, tree_lib.Tree)
    assert t.depth() <= 6
    assert t.num_nodes() >= 3
    txt = trained.print_tree(0)
    assert ">" in txt


def test_describe(trained):
    d = trained.describe()
    assert "GRADIENT_BOOSTED_TREES" in d
    assert "label" in d


def test_rf_save_load(tmp_path, binary_data):
    m = ydf.RandomForestLearner(label="label", num_trees=10).train(
        binary_data)
    p = str(tmp_path / "rf")
    m.save(p)
    m2 = ydf.load_model(p)
    assert isinstance(m2, ydf.RandomForestModel)
    np.testing.assert_allclose(m.predict(binary_data),
                               m2.predict(binary_data), rtol=1e-6, atol=1e-7)


def test_if_save_load(tmp_path):
    rng = np.random.RandomState(0)
    d = {"a": rng.randn(500).astype(np.float32)}
    m = ydf.IsolationForestLearner(num_trees=20).train(d)
    p = str(tmp_path / "if")
    m.save(p)
    m2 = ydf.load_model(p)
    assert isinstance(m2, ydf.IsolationForestModel)
    np.testing.assert_allclose(m.predict(d), m2.predict(d), rtol=1e-6,
                               atol=1e-7)


def test_evaluation_str(trained, binary_data):
    ev = trained.evaluate(binary_data)
    s = str(ev)
    assert "accuracy" in s
    assert ev._repr_html_()
