"""Deep tabular learners (PYDF ydf/deep/ analogue in PyTorch-ROCm)."""
import numpy as np
import pytest

import ydf_amd as ydf


@pytest.fixture(scope="module")
def clf_data():
    return ydf.generate_synthetic_dataset(
        num_examples=4000, num_numerical=5, num_categorical=2, seed=7)


def test_mlp_classification(clf_data, tmp_path):
    m = ydf.MultiLayerPerceptronLearner(
        label="LABEL", num_epochs=30, num_layers=3,
        layer_size=64).train(clf_data)
    ev = m.evaluate(clf_data)
    assert ev.auc > 0.78
    p = m.predict(clf_data)
    assert 0.0 <= p.min() and p.max() <= 1.0
    m.save(str(tmp_path / "mlp"))
    m2 = ydf.load_model(str(tmp_path / "mlp"))
    np.testing.assert_allclose(p, m2.predict(clf_data), atol=1e-6)
    assert m2.name() == "MLP"


def test_mlp_regression():
    d = ydf.generate_synthetic_dataset(num_examples=3000,
                                       task="regression", seed=8)
    m = ydf.MultiLayerPerceptronLearner(
        label="LABEL", task=ydf.Task.REGRESSION, num_epochs=30,
        num_layers=3, layer_size=64).train(d)
    assert m.evaluate(d).rmse < 0.5 * float(np.std(d["LABEL"]))


def test_mlp_multiclass():
    d = ydf.generate_synthetic_dataset(num_examples=3000, num_classes=3,
                                       seed=9)
    m = ydf.MultiLayerPerceptronLearner(
        label="LABEL", num_epochs=25, num_layers=2,
        layer_size=48).train(d)
    p = m.predict(d)
    assert p.shape[1] == 3
    np.testing.assert_allclose(p.sum(1), 1.0, atol=1e-4)
    assert m.evaluate(d).accuracy > 0.5


def test_tabular_transformer(clf_data):
    m = ydf.TabularTransformerLearner(
        label="LABEL", num_epochs=12, num_layers=2,
        token_dim=24).train(clf_data)
    assert m.evaluate(clf_data).auc > 0.75


def test_deep_early_stopping(clf_data):
    m = ydf.MultiLayerPerceptronLearner(
        label="LABEL", num_epochs=500, num_layers=2, layer_size=32,
        early_stopping_epoch_patience=3).train(clf_data)
    # patience must have cut the run well before 500 epochs
    assert len(m.training_logs) < 200
