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


def test_tree_builder_roundtrip():
    """build_model_from_trees (reference TreeBuilder analogue): trees
    extracted from a trained model rebuild into a bit-identical model;
    hand-built trees serve through the same kernels."""
    import ydf_amd as ydf

    d = ydf.generate_synthetic_dataset(num_examples=2000, num_numerical=4,
                                       num_categorical=1, seed=17)
    m = ydf.GradientBoostedTreesLearner(label="LABEL", num_trees=8,
                                        validation_ratio=0).train(d)
    trees = [ydf.extract_tree(m.forest, t) for t in range(m.num_trees())]
    m2 = ydf.build_model_from_trees(
        trees, m.dataspec, task=m.task(),
        init_predictions=m.init_predictions, activation=m.activation)
    m2.label_classes = m.label_classes
    np.testing.assert_array_equal(m.predict(d, device="cpu"),
                                  m2.predict(d, device="cpu"))
    t = ydf.Tree(root=ydf.NonLeaf(feature=0, threshold=0.5,
                                  neg_child=ydf.Leaf(1.0),
                                  pos_child=ydf.Leaf(2.0)))
    hm = ydf.build_model_from_trees([t], m.dataspec)
    x = {c.name: np.zeros(3, dtype=np.float32)
         for c in m.dataspec.feature_columns}
    x[m.dataspec.feature_columns[0].name] = np.array([0.0, 1.0, 0.4],
                                                     np.float32)
    np.testing.assert_allclose(hm.predict(x, device="cpu"),
                               [1.0, 2.0, 1.0])


def test_property_random_tree_roundtrip():
    """Property test: randomly-built trees (numerical + categorical-mask
    + oblique conditions) round-trip through predict/save/load/serialize
    bit-exactly."""
    import tempfile

    import ydf_amd as ydf
    from ydf_amd.dataset.dataspec import (ColumnSpec, DataSpecification,
                                          Semantic, Task)

    rng = np.random.RandomState(77)
    F = 6
    cols = [ColumnSpec(name=f"f{i}", semantic=Semantic.NUMERICAL)
            for i in range(F)]
    cols.append(ColumnSpec(name="y", semantic=Semantic.NUMERICAL))
    spec = DataSpecification(columns=cols, label="y")

    def random_tree(depth):
        if depth == 0 or rng.rand() < 0.3:
            return ydf.Leaf(float(rng.randn()))
        kind = rng.rand()
        if kind < 0.2:
            attrs = rng.choice(F, 2, replace=False)
            return ydf.NonLeaf(
                feature=int(attrs[0]), threshold=float(rng.randn()),
                neg_child=random_tree(depth - 1),
                pos_child=random_tree(depth - 1),
                oblique=(tuple(int(a) for a in attrs),
                         tuple(float(w) for w in rng.randn(2))))
        return ydf.NonLeaf(
            feature=int(rng.randint(F)), threshold=float(rng.randn()),
            neg_child=random_tree(depth - 1),
            pos_child=random_tree(depth - 1))

    for trial in range(5):
        trees = [ydf.Tree(root=random_tree(4)) for _ in range(7)]
        m = ydf.build_model_from_trees(trees, spec, task=ydf.Task.REGRESSION)
        d = {f"f{i}": rng.randn(500).astype(np.float32) for i in range(F)}
        p1 = m.predict(d, device="cpu")
        assert np.isfinite(p1).all()
        with tempfile.TemporaryDirectory() as td:
            m.save(td)
            np.testing.assert_array_equal(
                p1, ydf.load_model(td).predict(d, device="cpu"))
        m3 = ydf.deserialize_model(ydf.serialize_model(m))
        np.testing.assert_array_equal(p1, m3.predict(d, device="cpu"))
        # extraction round-trips the random structure too
        back = [ydf.extract_tree(m.forest, t) for t in range(7)]
        m4 = ydf.build_model_from_trees(back, spec,
                                        task=ydf.Task.REGRESSION)
        np.testing.assert_array_equal(p1, m4.predict(d, device="cpu"))


def test_pydf_method_surface(binary_data, tmp_path):
    """Thin PYDF model-method delegates: predict_class, serialize,
    to_cpp/to_standalone_cc/to_standalone_java, col idxs, logs."""
    import ydf_amd as ydf

    m = ydf.GradientBoostedTreesLearner(label="label", num_trees=5,
                                        validation_ratio=0).train(
        binary_data)
    cls = m.predict_class(binary_data)
    assert set(cls) <= set(m.label_classes)
    acc = (cls == binary_data["label"]).mean()
    assert acc > 0.9
    m2 = ydf.deserialize_model(m.serialize())
    np.testing.assert_allclose(m.predict(binary_data),
                               m2.predict(binary_data), rtol=1e-6)
    assert "predict" in m.to_cpp("k")
    assert "while ((fi" in m.to_standalone_cc()
    assert "class YdfModel" in m.to_standalone_java()
    assert m.input_features_col_idxs() == [0, 1, 2]
    assert m.hyperparameter_optimizer_logs() is None
    m.set_feature_selection_logs({"selected": ["x1"]})
    assert m.feature_selection_logs()["selected"] == ["x1"]
    with pytest.raises(ImportError):
        m.to_tensorflow_function()


def test_tree_editing(trained, binary_data):
    """set_tree/add_tree/remove_tree (reference PYDF
    decision_forest_model.py:148-173): edits round-trip through the
    flat forest and are visible in predictions."""
    n0 = trained.num_trees()
    p0 = trained.predict(binary_data)

    # replace tree 0 with a single constant leaf -> predictions move
    t0 = trained.get_tree(0)
    trained.set_tree(0, tree_lib.Tree(root=tree_lib.Leaf(value=0.0)))
    assert trained.num_trees() == n0
    p1 = trained.predict(binary_data)
    assert np.abs(p1 - p0).max() > 1e-6
    # restore the original tree -> predictions restored exactly
    trained.set_tree(0, t0)
    np.testing.assert_allclose(trained.predict(binary_data), p0,
                               rtol=1e-6, atol=1e-7)

    # add_tree / remove_tree keep the rest intact
    trained.add_tree(tree_lib.Tree(root=tree_lib.Leaf(value=0.25)))
    assert trained.num_trees() == n0 + 1
    p2 = trained.predict(binary_data)
    assert np.abs(p2 - p0).max() > 1e-3
    trained.remove_tree(n0)
    assert trained.num_trees() == n0
    np.testing.assert_allclose(trained.predict(binary_data), p0,
                               rtol=1e-6, atol=1e-7)


def test_tree_edit_preserves_categorical_and_na(binary_data):
    """Edit round-trip on a model with categorical masks and NA
    routing: untouched trees must serve identically after a rebuild."""
    data = dict(binary_data)
    rng = np.random.RandomState(5)
    x = data["x1"].astype(np.float32).copy()
    x[rng.rand(len(x)) < 0.2] = np.nan
    data["x1"] = x
    data["cat"] = rng.choice(["a", "b", "c", "d", "e"], len(x))
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=12, validation_ratio=0.0,
        allow_na_conditions=True).train(data)
    p0 = m.predict(data)
    t_last = m.get_tree(m.num_trees() - 1)
    m.set_tree(m.num_trees() - 1, t_last)  # identity edit
    np.testing.assert_allclose(m.predict(data), p0, rtol=1e-6,
                               atol=1e-7)


def test_plot_tree(trained, tmp_path):
    plot = trained.plot_tree(0, max_depth=8)
    html = plot.html()
    assert html.startswith("<svg") and html.endswith("</svg>")
    assert "value=" in html
    # depth-pruned rendering still works
    assert "<svg" in trained.plot_tree(0, max_depth=2).html()
    assert plot._repr_html_() == html
    f = str(tmp_path / "tree.svg")
    plot.to_file(f)
    assert "<svg" in open(f).read()


def test_distance_proximity(binary_data):
    """model.distance (reference PYDF decision_forest_model.py:196):
    tree-ensemble proximity — 0 on the diagonal, in [0,1], symmetric
    for one dataset, and near-duplicate examples are closer than
    random pairs."""
    m = ydf.RandomForestLearner(label="label", num_trees=30,
                                max_depth=10).train(binary_data)
    sub = {k: v[:200] for k, v in binary_data.items()}
    d = m.distance(sub)
    assert d.shape == (200, 200)
    np.testing.assert_allclose(np.diag(d), 0.0, atol=1e-7)
    assert d.min() >= 0.0 and d.max() <= 1.0
    np.testing.assert_allclose(d, d.T, atol=1e-7)
    # a tiny perturbation of an example stays closer than average
    pert = {k: (v[:50].astype(np.float32) + 1e-4
                if v.dtype.kind == "f" else v[:50])
            for k, v in binary_data.items()}
    dp = m.distance({k: v[:50] for k, v in binary_data.items()}, pert)
    assert np.diag(dp).mean() < d.mean() * 0.2

    # leaf_indices shape/type
    li = m.leaf_indices(sub)
    assert li.shape == (200, m.num_trees()) and li.dtype == np.int32


def test_reference_api_compat_surface(trained, binary_data):
    """Reference-API compatibility: label_classes/metadata/
    training_logs answer both attribute and method call forms; models
    pickle; predict_leaves/iter_trees/set_data_spec/set_node_format
    exist (PYDF generic_model.py / decision_forest_model.py)."""
    import pickle

    assert trained.label_classes == trained.label_classes()
    assert trained.metadata == trained.metadata()
    blob = pickle.dumps(trained)
    m2 = pickle.loads(blob)
    np.testing.assert_allclose(trained.predict(binary_data),
                               m2.predict(binary_data), rtol=1e-6)
    leaves = trained.predict_leaves(
        {k: v[:64] for k, v in binary_data.items()})
    assert leaves.shape == (64, trained.num_trees())
    assert sum(1 for _ in trained.iter_trees()) == trained.num_trees()
    trained.set_node_format("BLOB_SEQUENCE")
    with pytest.raises(ValueError):
        trained.set_node_format("TFE_RECORDIO")
    trained.set_data_spec(trained.data_spec())


def test_describe_formats(trained):
    """describe() output_format auto/text/html/notebook +
    full_details (reference generic_model.py:277)."""
    t = trained.describe()  # auto -> text
    assert "GRADIENT_BOOSTED_TREES" in t
    assert trained.describe("text") == t
    h = trained.describe("html")
    assert "<table>" in h
    nb = trained.describe("notebook")
    assert hasattr(nb, "_repr_html_") and "<table>" in nb._repr_html_()
    fd = trained.describe("text", full_details=True)
    assert "tree 0:" in fd and len(fd) > len(t)


def test_family_model_reference_methods(trained, binary_data):
    """Family-specific reference methods: GBT num_trees_per_iteration /
    output_logits / set_initial_predictions / validation_evaluation /
    early_stopping_triggered; RF out_of_bag_evaluations +
    winner_takes_all; IF num_examples_per_tree; activation duality."""
    assert trained.num_trees_per_iteration() == 1
    assert trained.activation == trained.activation()
    assert trained.output_logits() is False
    p0 = trained.predict(binary_data)
    trained.set_output_logits(True)
    logits = trained.predict(binary_data)
    np.testing.assert_allclose(1 / (1 + np.exp(-logits)), p0, atol=1e-5)
    trained.set_output_logits(False)
    bias0 = list(trained.init_predictions)
    trained.set_initial_predictions([float(bias0[0]) + 1.0])
    p1 = trained.predict(binary_data)
    assert (p1 >= p0 - 1e-6).all() and p1.mean() > p0.mean()
    trained.set_initial_predictions(bias0)
    assert trained.early_stopping_triggered() in (True, False, None)

    mrf = ydf.RandomForestLearner(label="label", num_trees=8).train(
        binary_data)
    assert mrf.out_of_bag_evaluations() is not None
    assert isinstance(mrf.winner_takes_all(), bool)
    m_nw = ydf.RandomForestLearner(label="label", num_trees=8,
                                   winner_take_all=False).train(
        binary_data)
    assert m_nw.winner_takes_all() is False

    rng = np.random.RandomState(0)
    mif = ydf.IsolationForestLearner(num_trees=10).train(
        {"a": rng.randn(500).astype(np.float32)})
    assert mif.num_examples_per_tree == 256
    assert mif.num_examples_per_tree() == 256


def test_training_log_entry_attr_access(binary_data):
    """training_logs entries answer the reference attribute form
    (entry.iteration, entry.evaluation) and the dict form."""
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=20, validation_ratio=0.2).train(
        binary_data)
    logs = m.training_logs()
    assert logs and logs[-1].iteration == logs[-1]["iteration"]
    ev = logs[-1].evaluation
    assert ev.loss == logs[-1]["valid_loss"]


def test_leaf_indices_reconstruct_predictions(binary_data):
    """Shapley-style consistency: summing leaf values at
    leaf_indices(+init, scale, activation) must reproduce predict()
    exactly — pins the walker against the serving kernels across
    condition types (numerical + categorical masks)."""
    data = dict(binary_data)
    rng = np.random.RandomState(3)
    data["cat"] = rng.choice(["u", "v", "w", "x"], len(data["x1"]))
    m = ydf.GradientBoostedTreesLearner(
        label="label", num_trees=25, validation_ratio=0.0).train(data)
    li = m.leaf_indices(data)
    margins = (m.init_predictions[0]
               + m.forest.thr[li].astype(np.float64).sum(axis=1)
               * m._leaf_scale())
    prob = 1.0 / (1.0 + np.exp(-margins))
    np.testing.assert_allclose(prob, m.predict(data), atol=2e-6)
