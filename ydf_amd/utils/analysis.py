"""Model analysis: variable importances, partial dependence, permutation
importance.

Capability analogue of the reference's utils/model_analysis.{h,cc} (Analyse
PDP/CEP reports), utils/partial_dependence_plot.*, and
utils/feature_importance.* (permutation importances,
random_forest.cc:1411-1477).
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional, Tuple

import numpy as np

from ydf_amd.dataset.dataspec import Semantic, Task


@dataclasses.dataclass
class PartialDependence:
    feature: str
    grid: np.ndarray           # grid values (numerical) or category ids
    mean_prediction: np.ndarray
    is_categorical: bool = False
    categories: Optional[List[str]] = None


@dataclasses.dataclass
class ConditionalExpectation:
    """Mean prediction and mean label over ACTUAL examples grouped by
    feature value (reference CEP plots, utils/model_analysis.cc)."""

    feature: str
    grid: np.ndarray
    mean_prediction: np.ndarray
    mean_label: np.ndarray
    counts: np.ndarray
    is_categorical: bool = False
    categories: Optional[List[str]] = None


@dataclasses.dataclass
class Analysis:
    """Analysis report (mirrors ydf model.analyze() output)."""

    variable_importances: Dict[str, List[Tuple[float, str]]]
    partial_dependences: List[PartialDependence]
    conditional_expectations: List[ConditionalExpectation] = \
        dataclasses.field(default_factory=list)

    def to_text(self) -> str:
        out = []
        for name, vi in self.variable_importances.items():
            out.append(f"Variable importance: {name}")
            for rank, (score, feat) in enumerate(vi, 1):
                out.append(f"  {rank:3d}. {feat:30s} {score:.6g}")
            out.append("")
        for pd in self.partial_dependences[:10]:
            out.append(f"PDP {pd.feature}: "
                       + " ".join(f"{v:.3g}" for v in
                                  pd.mean_prediction[:8]))
        return "\n".join(out)

    def __str__(self) -> str:
        return self.to_text()

    def _repr_html_(self) -> str:
        """Full HTML report: VI bar charts + per-feature PDP curves
        (reference utils/model_analysis.h:36-89 CreateHtmlReport /
        utils/plot.* — rendered as dependency-free inline SVG instead
        of plotly)."""
        rows = []
        for name, vi in self.variable_importances.items():
            body = "".join(
                f"<tr><td>{f}</td><td>{s:.6g}</td></tr>" for s, f in vi)
            rows.append(f"<h3>{name}</h3>{_svg_bar_chart(vi)}"
                        f"<details><summary>table</summary>"
                        f"<table><tr><th>feature</th>"
                        f"<th>score</th></tr>{body}</table></details>")
        if self.conditional_expectations:
            rows.append("<h3>Conditional expectation (on data)</h3>"
                        "<div style='display:flex;flex-wrap:wrap'>")
            for ce in self.conditional_expectations:
                pd_like = PartialDependence(
                    feature=ce.feature, grid=ce.grid,
                    mean_prediction=ce.mean_prediction,
                    is_categorical=ce.is_categorical,
                    categories=ce.categories)
                rows.append(
                    f"<div style='margin:4px'>"
                    f"<b style='font-size:12px'>{ce.feature}</b><br/>"
                    + _svg_pdp(pd_like) + "</div>")
            rows.append("</div>")
        if self.partial_dependences:
            rows.append("<h3>Partial dependence</h3>"
                        "<div style='display:flex;flex-wrap:wrap'>")
            for pd in self.partial_dependences:
                rows.append(
                    f"<div style='margin:4px'>"
                    f"<b style='font-size:12px'>{pd.feature}</b><br/>"
                    + _svg_pdp(pd) + "</div>")
            rows.append("</div>")
        return "".join(rows)


def _svg_bar_chart(vi: List[Tuple[float, str]], width: int = 420,
                   bar_h: int = 14, top: int = 12) -> str:
    """Horizontal bar chart of the top variable importances."""
    items = vi[:top]
    if not items:
        return ""
    mx = max(abs(s) for s, _ in items) or 1.0
    h = len(items) * (bar_h + 3) + 4
    parts = [f'<svg width="{width}" height="{h}" '
             'xmlns="http://www.w3.org/2000/svg" '
             'style="font:10px sans-serif">']
    for i, (s, f) in enumerate(items):
        y = 2 + i * (bar_h + 3)
        w = max(1.0, abs(s) / mx * (width - 190))
        parts.append(
            f'<rect x="150" y="{y}" width="{w:.1f}" height="{bar_h}" '
            'fill="#4a7abc"/>')
        parts.append(f'<text x="146" y="{y + bar_h - 3}" '
                     f'text-anchor="end">{f[:24]}</text>')
        parts.append(f'<text x="{152 + w:.1f}" y="{y + bar_h - 3}">'
                     f'{s:.4g}</text>')
    parts.append("</svg>")
    return "".join(parts)


def _svg_pdp(pd: PartialDependence, width: int = 220,
             height: int = 120) -> str:
    """One PDP panel: line plot (numerical) or bars (categorical)."""
    ys = np.asarray(pd.mean_prediction, dtype=np.float64)
    if ys.size == 0:
        return ""
    lo, hi = float(ys.min()), float(ys.max())
    span = (hi - lo) or 1.0
    pad = 14
    ph = height - 2 * pad
    pw = width - 2 * pad

    def sy(v):
        return pad + ph - (v - lo) / span * ph

    parts = [f'<svg width="{width}" height="{height}" '
             'xmlns="http://www.w3.org/2000/svg" '
             'style="font:9px sans-serif;background:#fafafa">']
    parts.append(f'<text x="2" y="10">{hi:.3g}</text>')
    parts.append(f'<text x="2" y="{height - 2}">{lo:.3g}</text>')
    n = ys.size
    if pd.is_categorical:
        bw = pw / max(n, 1)
        for i, v in enumerate(ys):
            x = pad + i * bw
            parts.append(
                f'<rect x="{x:.1f}" y="{sy(v):.1f}" '
                f'width="{max(bw - 2, 1):.1f}" '
                f'height="{pad + ph - sy(v):.1f}" fill="#4a7abc"/>')
    else:
        pts = " ".join(
            f"{pad + i / max(n - 1, 1) * pw:.1f},{sy(v):.1f}"
            for i, v in enumerate(ys))
        parts.append(f'<polyline points="{pts}" fill="none" '
                     'stroke="#c0392b" stroke-width="1.5"/>')
        gx = np.asarray(pd.grid, dtype=np.float64)
        if gx.size:
            parts.append(f'<text x="{pad}" y="{height - 2}" '
                         f'text-anchor="start"></text>')
            parts.append(
                f'<text x="{width - 2}" y="{height - 2}" '
                f'text-anchor="end">{gx[-1]:.3g}</text>')
            parts.append(
                f'<text x="{pad}" y="{height - 2}">{gx[0]:.3g}</text>')
    parts.append("</svg>")
    return "".join(parts)


def structure_importances(model) -> Dict[str, List[Tuple[float, str]]]:
    """Importances from the forest structure: number of nodes per feature
    and number-of-times-root (reference NUM_NODES / NUM_AS_ROOT)."""
    names = model.input_feature_names()
    forest = model.forest
    num_nodes = np.zeros(len(names), dtype=np.int64)
    used = forest.feat[forest.feat >= 0]
    np.add.at(num_nodes, used, 1)
    num_root = np.zeros(len(names), dtype=np.int64)
    root_feats = forest.feat[forest.roots]
    np.add.at(num_root, root_feats[root_feats >= 0], 1)

    def ranked(scores):
        order = np.argsort(-scores, kind="stable")
        return [(float(scores[i]), names[i]) for i in order if scores[i] > 0]

    out = {
        "NUM_NODES": ranked(num_nodes.astype(np.float64)),
        "NUM_AS_ROOT": ranked(num_root.astype(np.float64)),
    }
    gains = model.metadata.get("feature_gains") if model.metadata else None
    if gains:
        out["SUM_SCORE"] = sorted(
            ((float(v), k) for k, v in gains.items() if v > 0), reverse=True)
    oob_vi = model.metadata.get("oob_permutation_importances") \
        if model.metadata else None
    if oob_vi:
        for metric, ranked_list in oob_vi.items():
            out[metric] = [(float(s), n) for s, n in ranked_list]
    pvi = model.metadata.get("permutation_importances") \
        if model.metadata else None
    if pvi:
        for metric, ranked_list in pvi.items():
            out[metric] = [(float(s), n) for s, n in ranked_list]
    return out


def permutation_importances(model, data, labels: np.ndarray,
                            num_repetitions: int = 1,
                            seed: int = 1234,
                            device=None) -> List[Tuple[float, str]]:
    """Mean metric drop when a feature column is shuffled (reference
    MEAN_DECREASE_IN_ACCURACY / permutation variable importances)."""
    from ydf_amd.metric.metric import accuracy, rmse

    X = model._encode_features(data).copy()
    names = model.input_feature_names()
    rng = np.random.RandomState(seed)

    def score(Xm):
        import torch

        dev = (torch.device(device) if device is not None
               else (torch.device("cuda") if torch.cuda.is_available()
                     else torch.device("cpu")))
        Xt = torch.from_numpy(np.ascontiguousarray(Xm)).to(dev)
        m = model.predict_margin(Xt)
        p = model._apply_activation(m).cpu().numpy()
        if model.task() == Task.CLASSIFICATION:
            if p.ndim == 1:
                return accuracy(labels.astype(np.int64),
                                (p >= 0.5).astype(np.int64))
            return accuracy(labels.astype(np.int64), p.argmax(axis=1))
        return -rmse(labels, p)

    base = score(X)
    drops = []
    for fi, name in enumerate(names):
        drop = 0.0
        saved = X[fi].copy()
        for _ in range(num_repetitions):
            X[fi] = saved[rng.permutation(X.shape[1])]
            drop += base - score(X)
        X[fi] = saved
        drops.append((drop / num_repetitions, name))
    drops.sort(reverse=True)
    return drops


def partial_dependences(model, data, features: Optional[List[str]] = None,
                        num_grid_points: int = 20, max_examples: int = 5000,
                        device=None) -> List[PartialDependence]:
    import torch

    X = model._encode_features(data)
    if X.shape[1] > max_examples:
        idx = np.random.RandomState(0).choice(X.shape[1], max_examples,
                                              replace=False)
        X = X[:, idx]
    X = X.copy()
    specs = model.dataspec.feature_columns
    names = model.input_feature_names()
    wanted = set(features) if features else None
    dev = (torch.device(device) if device is not None
           else (torch.device("cuda") if torch.cuda.is_available()
                 else torch.device("cpu")))

    def mean_pred(Xm):
        Xt = torch.from_numpy(np.ascontiguousarray(Xm)).to(dev)
        p = model._apply_activation(model.predict_margin(Xt)).cpu().numpy()
        if p.ndim == 2:  # multi-class: track P(class index 1) like binary
            p = p[:, min(1, p.shape[1] - 1)]
        return float(p.mean())

    out = []
    for fi, spec in enumerate(specs):
        if wanted is not None and spec.name not in wanted:
            continue
        saved = X[fi].copy()
        if spec.semantic == Semantic.CATEGORICAL:
            cats = list(range(min(spec.vocab_size, 16)))
            means = []
            for c in cats:
                X[fi] = float(c)
                means.append(mean_pred(X))
            out.append(PartialDependence(
                feature=spec.name, grid=np.asarray(cats, dtype=np.float32),
                mean_prediction=np.asarray(means, dtype=np.float32),
                is_categorical=True,
                categories=[spec.vocab[c] for c in cats]))
        else:
            qs = np.linspace(0.02, 0.98, num_grid_points)
            grid = np.quantile(saved, qs).astype(np.float32)
            means = []
            for v in grid:
                X[fi] = v
                means.append(mean_pred(X))
            out.append(PartialDependence(
                feature=spec.name, grid=grid,
                mean_prediction=np.asarray(means, dtype=np.float32)))
        X[fi] = saved
    return out


def conditional_expectations(model, data, labels=None,
                             num_bins: int = 16,
                             max_examples: int = 20000,
                             device=None):
    """Mean prediction (and mean label when given) over actual
    examples bucketed by each feature's value — the reference's CEP
    companion to PDP (utils/model_analysis.cc)."""
    import torch

    X = model._encode_features(data)
    if labels is not None:
        labels = np.asarray(labels, dtype=np.float64)
    if X.shape[1] > max_examples:
        idx = np.random.RandomState(0).choice(X.shape[1], max_examples,
                                              replace=False)
        X = X[:, idx]
        if labels is not None:
            labels = labels[idx]
    dev = (torch.device(device) if device is not None
           else (torch.device("cuda") if torch.cuda.is_available()
                 else torch.device("cpu")))
    Xt = torch.from_numpy(np.ascontiguousarray(X)).to(dev)
    p = model._apply_activation(model.predict_margin(Xt)).cpu().numpy()
    if p.ndim == 2:
        p = p[:, min(1, p.shape[1] - 1)]
    p = p.astype(np.float64)
    out = []
    for fi, spec in enumerate(model.dataspec.feature_columns):
        vals = X[fi]
        if spec.semantic == Semantic.CATEGORICAL:
            K = min(spec.vocab_size, 16)
            codes = np.clip(vals.astype(np.int64), 0, K - 1)
            cnt = np.bincount(codes, minlength=K)
            mp = np.bincount(codes, weights=p, minlength=K) \
                / np.maximum(cnt, 1)
            ml = (np.bincount(codes, weights=labels, minlength=K)
                  / np.maximum(cnt, 1)) if labels is not None \
                else np.full(K, np.nan)
            out.append(ConditionalExpectation(
                feature=spec.name,
                grid=np.arange(K, dtype=np.float32),
                mean_prediction=mp.astype(np.float32),
                mean_label=ml.astype(np.float32),
                counts=cnt.astype(np.int64), is_categorical=True,
                categories=[spec.vocab[c] for c in range(K)]))
        else:
            edges = np.quantile(vals, np.linspace(0, 1, num_bins + 1))
            edges = np.unique(edges)
            if len(edges) < 2:
                continue
            codes = np.clip(np.searchsorted(edges, vals, side="right")
                            - 1, 0, len(edges) - 2)
            K = len(edges) - 1
            cnt = np.bincount(codes, minlength=K)
            mp = np.bincount(codes, weights=p, minlength=K) \
                / np.maximum(cnt, 1)
            ml = (np.bincount(codes, weights=labels, minlength=K)
                  / np.maximum(cnt, 1)) if labels is not None \
                else np.full(K, np.nan)
            centers = (edges[:-1] + edges[1:]) / 2
            out.append(ConditionalExpectation(
                feature=spec.name,
                grid=centers.astype(np.float32),
                mean_prediction=mp.astype(np.float32),
                mean_label=ml.astype(np.float32),
                counts=cnt.astype(np.int64)))
    return out


def analyze(model, data, labels: Optional[np.ndarray] = None,
            permutation_variable_importance: bool = True,
            partial_dependence: bool = True,
            conditional_expectation: bool = True,
            shap_values: bool = True,
            features: Optional[List[str]] = None,
            num_grid_points: int = 20,
            permutation_rounds: int = 1, device=None) -> Analysis:
    vi = structure_importances(model)
    if permutation_variable_importance and labels is not None:
        key = ("MEAN_DECREASE_IN_ACCURACY"
               if model.task() == Task.CLASSIFICATION
               else "MEAN_INCREASE_IN_RMSE")
        vi[key] = permutation_importances(
            model, data, labels,
            num_repetitions=max(1, permutation_rounds), device=device)
    if shap_values:
        try:
            phi = model.predict_shap(data)
            shap_vi = sorted(
                ((float(np.abs(v).mean()), k)
                 for k, v in phi.items() if k != "__BIAS__"),
                reverse=True)
            vi["MEAN_ABS_SHAP"] = shap_vi
        except (NotImplementedError, ValueError):
            pass  # multi-output / coverless models: no SHAP summary
    pdps = []
    ceps = []
    if partial_dependence:
        pdps = partial_dependences(model, data, features=features,
                                   num_grid_points=num_grid_points,
                                   device=device)
    if conditional_expectation:
        ceps = conditional_expectations(model, data, labels=labels,
                                        device=device)
    return Analysis(variable_importances=vi, partial_dependences=pdps,
                    conditional_expectations=ceps)
