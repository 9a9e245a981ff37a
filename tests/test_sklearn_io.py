"""sklearn model import parity (capability analogue of ydf.from_sklearn,
port/python/ydf/model/export_sklearn.py): converted models must reproduce
sklearn's own predictions through our serving kernels."""
import numpy as np
import pytest

import ydf_amd as ydf

sklearn = pytest.importorskip("sklearn")

from sklearn.ensemble import (GradientBoostingClassifier,  # noqa: E402
                              GradientBoostingRegressor, IsolationForest,
                              RandomForestClassifier, RandomForestRegressor)
from sklearn.tree import DecisionTreeClassifier  # noqa: E402


@pytest.fixture(scope="module")
def Xy():
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 5).astype(np.float32)
    return X, {f"features[{i}]": X[:, i] for i in range(5)}


def test_rf_classifier_binary(Xy):
    X, d = Xy
    y = (X[:, 0] + 2 * X[:, 1] > 0).astype(int)
    sk = RandomForestClassifier(n_estimators=10, random_state=0).fit(X, y)
    m = ydf.from_sklearn(sk)
    np.testing.assert_allclose(m.predict(d), sk.predict_proba(X)[:, 1],
                               atol=1e-5)
    assert m.name() == "RANDOM_FOREST"


def test_rf_classifier_multiclass(Xy):
    X, d = Xy
    y = (X[:, 0] + 2 * X[:, 1] > 0).astype(int) + (X[:, 2] > 1).astype(int)
    sk = RandomForestClassifier(n_estimators=8, random_state=0).fit(X, y)
    m = ydf.from_sklearn(sk)
    np.testing.assert_allclose(m.predict(d), sk.predict_proba(X), atol=1e-5)


def test_rf_regressor(Xy):
    X, d = Xy
    y = (X[:, 0] * 2 - X[:, 1]).astype(np.float32)
    sk = RandomForestRegressor(n_estimators=8, random_state=0).fit(X, y)
    m = ydf.from_sklearn(sk)
    np.testing.assert_allclose(m.predict(d), sk.predict(X), atol=1e-4)


def test_gbt_classifier_and_save_roundtrip(Xy, tmp_path):
    X, d = Xy
    y = (X[:, 0] + 2 * X[:, 1] > 0).astype(int)
    sk = GradientBoostingClassifier(n_estimators=20, random_state=0).fit(X,
                                                                         y)
    m = ydf.from_sklearn(sk)
    want = sk.predict_proba(X)[:, 1]
    np.testing.assert_allclose(m.predict(d), want, atol=1e-5)
    m.save(str(tmp_path / "m"))
    m2 = ydf.load_model(str(tmp_path / "m"))
    np.testing.assert_allclose(m2.predict(d), want, atol=1e-5)


def test_gbt_regressor(Xy):
    X, d = Xy
    y = (X[:, 0] * 2 - X[:, 1]).astype(np.float32)
    sk = GradientBoostingRegressor(n_estimators=20, random_state=0).fit(X, y)
    m = ydf.from_sklearn(sk)
    np.testing.assert_allclose(m.predict(d), sk.predict(X), atol=1e-4)


def test_isolation_forest(Xy):
    X, d = Xy
    sk = IsolationForest(n_estimators=20, random_state=0).fit(X)
    m = ydf.from_sklearn(sk)
    # sklearn's -score_samples IS 2^(-E[h]/c(n)), our anomaly score
    np.testing.assert_allclose(m.predict(d), -sk.score_samples(X),
                               atol=1e-3)


def test_decision_tree(Xy):
    X, d = Xy
    y = (X[:, 0] + 2 * X[:, 1] > 0).astype(int)
    sk = DecisionTreeClassifier(max_depth=5, random_state=0).fit(X, y)
    m = ydf.from_sklearn(sk)
    np.testing.assert_allclose(m.predict(d), sk.predict_proba(X)[:, 1],
                               atol=1e-5)
