"""Evaluation metrics (capability analogue of the reference metric layer,
yggdrasil_decision_forests/metric/metric.h:42-177: accuracy, AUC, PR-AUC,
log-loss, RMSE/MAE, NDCG/MRR, confusion matrix, closed-form + bootstrap
confidence intervals)."""
from __future__ import annotations

import dataclasses
from typing import Dict, Optional

import numpy as np


def accuracy(labels: np.ndarray, pred_classes: np.ndarray) -> float:
    return float((labels == pred_classes).mean()) if len(labels) else 0.0


def confusion_matrix(labels: np.ndarray, pred_classes: np.ndarray,
                     n_classes: int) -> np.ndarray:
    cm = np.zeros((n_classes, n_classes), dtype=np.int64)
    np.add.at(cm, (labels.astype(np.int64), pred_classes.astype(np.int64)), 1)
    return cm


def roc_curve(labels: np.ndarray, scores: np.ndarray):
    """(fpr, tpr, thresholds) swept over unique score cutoffs
    (descending), the reference's Roc building blocks."""
    labels = np.asarray(labels, dtype=bool)
    order = np.argsort(-scores, kind="mergesort")
    s = scores[order]
    y = labels[order]
    distinct = np.nonzero(np.diff(s))[0]
    idx = np.r_[distinct, len(s) - 1]
    tps = np.cumsum(y)[idx].astype(np.float64)
    fps = (idx + 1) - tps
    P = max(float(labels.sum()), 1.0)
    Nn = max(float(len(labels) - labels.sum()), 1.0)
    return fps / Nn, tps / P, s[idx]


def roc_auc(labels: np.ndarray, scores: np.ndarray) -> float:
    """Rank-based AUC (equivalent to the trapezoidal ROC integral used by
    the reference, metric.h:150)."""
    labels = np.asarray(labels, dtype=bool)
    n_pos = int(labels.sum())
    n_neg = len(labels) - n_pos
    if n_pos == 0 or n_neg == 0:
        return float("nan")
    order = np.argsort(scores, kind="mergesort")
    ranks = np.empty(len(scores), dtype=np.float64)
    ranks[order] = np.arange(1, len(scores) + 1)
    # average ranks for ties
    sorted_scores = scores[order]
    uniq, inv, cnt = np.unique(sorted_scores, return_inverse=True,
                               return_counts=True)
    cum = np.cumsum(cnt)
    avg_rank = (cum - (cnt - 1) / 2.0)
    ranks[order] = avg_rank[inv]
    sum_pos = ranks[labels].sum()
    return float((sum_pos - n_pos * (n_pos + 1) / 2.0) / (n_pos * n_neg))


def pr_auc(labels: np.ndarray, scores: np.ndarray) -> float:
    labels = np.asarray(labels, dtype=bool)
    order = np.argsort(-scores, kind="mergesort")
    tp = np.cumsum(labels[order])
    fp = np.cumsum(~labels[order])
    n_pos = int(labels.sum())
    if n_pos == 0:
        return float("nan")
    precision = tp / np.maximum(tp + fp, 1)
    recall = tp / n_pos
    # step-wise integration
    dr = np.diff(np.concatenate([[0.0], recall]))
    return float((precision * dr).sum())


def log_loss(labels: np.ndarray, probs: np.ndarray,
             eps: float = 1e-12) -> float:
    """labels: int class idx; probs: [N] (binary, P(class1)) or [N,C]."""
    if probs.ndim == 1:
        # float64 first: clipping float32 probs against 1-1e-12 rounds to 1.0
        p = np.clip(probs.astype(np.float64), eps, 1 - eps)
        y = labels.astype(np.float64)
        return float(-(y * np.log(p) + (1 - y) * np.log(1 - p)).mean())
    p = np.clip(probs[np.arange(len(labels)),
                      labels.astype(int)].astype(np.float64), eps, 1.0)
    return float(-np.log(p).mean())


def rmse(labels: np.ndarray, preds: np.ndarray) -> float:
    return float(np.sqrt(((labels - preds) ** 2).mean()))


def mae(labels: np.ndarray, preds: np.ndarray) -> float:
    return float(np.abs(labels - preds).mean())


def ndcg(labels: np.ndarray, scores: np.ndarray, groups: np.ndarray,
         truncation: int = 5) -> float:
    """Mean NDCG@truncation over ranking groups (reference metric/ranking)."""
    total, n_groups = 0.0, 0
    for g in np.unique(groups):
        m = groups == g
        rel = labels[m]
        sc = scores[m]
        if len(rel) == 0:
            continue
        order = np.argsort(-sc, kind="mergesort")
        gains = (2.0 ** rel[order][:truncation] - 1)
        discounts = 1.0 / np.log2(np.arange(2, len(gains) + 2))
        dcg = float((gains * discounts).sum())
        iorder = np.argsort(-rel, kind="mergesort")
        igains = (2.0 ** rel[iorder][:truncation] - 1)
        idcg = float((igains * discounts[:len(igains)]).sum())
        if idcg > 0:
            total += dcg / idcg
            n_groups += 1
    return total / n_groups if n_groups else float("nan")


def mrr(labels: np.ndarray, scores: np.ndarray, groups: np.ndarray,
        truncation: int = 5) -> float:
    """Mean reciprocal rank of the first relevant (label > 0) item
    within the truncation (reference metric/ranking_mrr.h)."""
    total, n_groups = 0.0, 0
    for g in np.unique(groups):
        m = groups == g
        rel = labels[m]
        if not (rel > 0).any():
            continue
        order = np.argsort(-scores[m], kind="mergesort")[:truncation]
        hit = np.nonzero(rel[order] > 0)[0]
        total += 1.0 / (hit[0] + 1) if len(hit) else 0.0
        n_groups += 1
    return total / n_groups if n_groups else float("nan")


def mean_average_precision(labels: np.ndarray, scores: np.ndarray,
                           groups: np.ndarray,
                           truncation: int = 5) -> float:
    """Ranking MAP@truncation: mean over groups of average precision
    with binary relevance label > 0 (reference ranking metric "map")."""
    total, n_groups = 0.0, 0
    for g in np.unique(groups):
        m = groups == g
        rel = labels[m] > 0
        if not rel.any():
            continue
        order = np.argsort(-scores[m], kind="mergesort")[:truncation]
        r = rel[order]
        hits = np.cumsum(r)
        prec_at_hit = hits[r] / (np.nonzero(r)[0] + 1)
        total += float(prec_at_hit.mean())
        n_groups += 1
    return total / n_groups if n_groups else float("nan")


def accuracy_confidence_interval(acc: float, n: int,
                                 level: float = 0.95):
    """Wilson score interval (closed form; reference
    metric.h ComputeXsLossConfidenceInterval family)."""
    if n == 0:
        return (float("nan"), float("nan"))
    from scipy.stats import norm

    z = norm.ppf(0.5 + level / 2.0)
    denom = 1.0 + z * z / n
    center = (acc + z * z / (2 * n)) / denom
    half = z * np.sqrt(acc * (1 - acc) / n + z * z / (4 * n * n)) / denom
    return (float(center - half), float(center + half))


def auc_confidence_interval(auc: float, n_pos: int, n_neg: int,
                            level: float = 0.95):
    """Hanley-McNeil closed-form AUC standard error."""
    if n_pos == 0 or n_neg == 0 or not np.isfinite(auc):
        return (float("nan"), float("nan"))
    from scipy.stats import norm

    q1 = auc / (2 - auc)
    q2 = 2 * auc * auc / (1 + auc)
    se = np.sqrt((auc * (1 - auc) + (n_pos - 1) * (q1 - auc * auc)
                  + (n_neg - 1) * (q2 - auc * auc)) / (n_pos * n_neg))
    z = norm.ppf(0.5 + level / 2.0)
    return (float(max(0.0, auc - z * se)), float(min(1.0, auc + z * se)))


def bootstrap_confidence_intervals(labels: np.ndarray,
                                   predictions: np.ndarray, task,
                                   n_samples: int = 500,
                                   level: float = 0.95,
                                   seed: int = 1234):
    """Percentile-bootstrap CIs for the task's main metrics (reference
    bootstrap CIs, metric.h:150-177). Returns {metric: (lo, hi)}."""
    from ydf_amd.dataset.dataspec import Task

    rng = np.random.RandomState(seed)
    n = len(labels)
    stats = {}
    for _ in range(n_samples):
        idx = rng.randint(0, n, n)
        ev = evaluate_predictions(predictions[idx], labels[idx], task)
        for k in ("accuracy", "auc", "rmse", "mae", "loss"):
            v = getattr(ev, k)
            if v is not None and np.isfinite(v):
                stats.setdefault(k, []).append(v)
    lo_q, hi_q = 100 * (0.5 - level / 2), 100 * (0.5 + level / 2)
    return {k: (float(np.percentile(v, lo_q)),
                float(np.percentile(v, hi_q)))
            for k, v in stats.items() if len(v) > 1}


@dataclasses.dataclass
class Characteristic:
    """Per-threshold binary-classification curve (mirrors PYDF
    metric.Characteristic): ROC points plus derived precision/recall
    arrays from the stored class counts. Also answers dict-style
    access (ch["fpr"]) for backward compatibility."""

    name: str
    fpr: np.ndarray
    tpr: np.ndarray
    thresholds: np.ndarray
    n_pos: int = 0
    n_neg: int = 0
    roc_auc: Optional[float] = None
    pr_auc: Optional[float] = None

    def __getitem__(self, k):
        return getattr(self, k)

    @property
    def recalls(self) -> np.ndarray:
        return self.tpr

    @property
    def false_positive_rates(self) -> np.ndarray:
        return self.fpr

    @property
    def precisions(self) -> np.ndarray:
        tp = self.tpr * self.n_pos
        fp = self.fpr * self.n_neg
        with np.errstate(invalid="ignore", divide="ignore"):
            return np.where(tp + fp > 0, tp / (tp + fp), 1.0)

    @property
    def accuracies(self) -> np.ndarray:
        tp = self.tpr * self.n_pos
        tn = self.n_neg - self.fpr * self.n_neg
        return (tp + tn) / max(self.n_pos + self.n_neg, 1)

    def precision_at_recall(self, recall: float) -> float:
        if recall <= 0.0:
            return 1.0
        m = self.recalls >= recall
        return float(self.precisions[m].max()) if m.any() else 0.0


@dataclasses.dataclass
class ConfusionMatrix:
    """Confusion matrix with class names (mirrors PYDF
    metric.ConfusionMatrix): rows = truth, cols = prediction."""

    classes: tuple
    matrix: np.ndarray

    def value(self, label, prediction) -> float:
        return float(self.matrix[self.classes.index(label),
                                 self.classes.index(prediction)])

    def __str__(self) -> str:
        head = "truth\\pred " + " ".join(str(c) for c in self.classes)
        rows = [f"{c} " + " ".join(str(int(v)) for v in self.matrix[i])
                for i, c in enumerate(self.classes)]
        return "\n".join([head] + rows)


@dataclasses.dataclass
class Evaluation:
    """Evaluation report (mirrors ydf.metric.Evaluation fields)."""

    num_examples: int = 0
    accuracy: Optional[float] = None
    loss: Optional[float] = None
    auc: Optional[float] = None
    pr_auc: Optional[float] = None
    rmse: Optional[float] = None
    mae: Optional[float] = None
    ndcg: Optional[float] = None
    mrr: Optional[float] = None
    auuc: Optional[float] = None
    qini: Optional[float] = None
    cindex: Optional[float] = None
    confusion: Optional[np.ndarray] = None
    # label class names (classification; feeds confusion_matrix)
    classes: Optional[tuple] = None
    num_examples_weighted: Optional[float] = None
    custom_metrics: Optional[Dict] = None
    # ranking mean average precision (reference ranking metric "map")
    map: Optional[float] = None
    # regression bootstrap 95% CI on RMSE (reference
    # bootstrap_rmse_*_bounds_95p)
    rmse_ci95_bootstrap: Optional[tuple] = None
    # per-threshold ROC points for binary classification (PYDF
    # evaluation.characteristics; reference metric Roc curves):
    # list of dicts {"name", "fpr", "tpr", "thresholds"}
    characteristics: Optional[list] = None
    # closed-form 95% confidence intervals (lo, hi)
    accuracy_ci95: Optional[tuple] = None
    auc_ci95: Optional[tuple] = None

    # -- derived binary-classification statistics (from the confusion
    # matrix; convention: rows = truth, cols = prediction, class 1 =
    # positive) ------------------------------------------------------
    def _binary_counts(self):
        cm = self.confusion
        if cm is None or cm.shape != (2, 2):
            return None
        tn, fp = float(cm[0, 0]), float(cm[0, 1])
        fn, tp = float(cm[1, 0]), float(cm[1, 1])
        return tp, fp, tn, fn

    @property
    def precision(self) -> Optional[float]:
        c = self._binary_counts()
        if c is None:
            return None
        tp, fp, _, _ = c
        return tp / (tp + fp) if tp + fp > 0 else float("nan")

    @property
    def recall(self) -> Optional[float]:
        c = self._binary_counts()
        if c is None:
            return None
        tp, _, _, fn = c
        return tp / (tp + fn) if tp + fn > 0 else float("nan")

    @property
    def f1(self) -> Optional[float]:
        p, r = self.precision, self.recall
        if p is None or r is None or p + r == 0:
            return None
        return 2 * p * r / (p + r)

    @property
    def false_positive_rate(self) -> Optional[float]:
        c = self._binary_counts()
        if c is None:
            return None
        _, fp, tn, _ = c
        return fp / (fp + tn) if fp + tn > 0 else float("nan")

    def to_dict(self) -> Dict:
        d = {"num_examples": self.num_examples}
        for k in ("accuracy", "loss", "auc", "pr_auc", "rmse", "mae",
                  "ndcg", "mrr", "auuc", "qini", "cindex",
                  "accuracy_ci95", "auc_ci95"):
            v = getattr(self, k)
            if v is not None:
                d[k] = v
        return d

    def __str__(self) -> str:
        parts = [f"num examples: {self.num_examples}"]
        for k in ("accuracy", "loss", "auc", "pr_auc", "rmse", "mae",
                  "ndcg", "mrr", "auuc", "qini", "cindex"):
            v = getattr(self, k)
            if v is not None:
                parts.append(f"{k}: {v:.6g}")
        for k in ("accuracy_ci95", "auc_ci95"):
            v = getattr(self, k)
            if v is not None:
                parts.append(f"{k}: [{v[0]:.6g}, {v[1]:.6g}]")
        return "\n".join(parts)

    @property
    def confusion_matrix(self):
        """ConfusionMatrix with class names (PYDF
        evaluation.confusion_matrix); None for non-classification."""
        if self.confusion is None:
            return None
        cls = self.classes or tuple(range(self.confusion.shape[0]))
        return ConfusionMatrix(classes=tuple(cls), matrix=self.confusion)

    def html(self) -> str:
        """HTML report (PYDF evaluation.html())."""
        return self._repr_html_()

    def _repr_html_(self) -> str:
        rows = "".join(
            f"<tr><td>{k}</td><td>{v:.6g}</td></tr>"
            if isinstance(v, float) else f"<tr><td>{k}</td><td>{v}</td></tr>"
            for k, v in self.to_dict().items())
        html = f"<table>{rows}</table>"
        if self.characteristics:
            # ROC curve panel (PYDF evaluation HTML shows the plotly
            # ROC; here dependency-free inline SVG)
            ch = self.characteristics[0]
            w = h = 180
            pad = 16
            pts = " ".join(
                f"{pad + x * (w - 2 * pad):.1f},"
                f"{h - pad - y * (h - 2 * pad):.1f}"
                for x, y in zip(ch["fpr"], ch["tpr"]))
            html += (
                f'<div><b style="font:11px sans-serif">ROC</b><br/>'
                f'<svg width="{w}" height="{h}" '
                'xmlns="http://www.w3.org/2000/svg" '
                'style="background:#fafafa">'
                f'<line x1="{pad}" y1="{h - pad}" x2="{w - pad}" '
                f'y2="{pad}" stroke="#ccc" stroke-dasharray="3"/>'
                f'<polyline points="{pts}" fill="none" stroke="#2a7" '
                'stroke-width="1.5"/></svg></div>')
        return html


def evaluate_predictions(predictions: np.ndarray, labels: np.ndarray,
                         task, n_classes: int = 2,
                         weights: Optional[np.ndarray] = None
                         ) -> Evaluation:
    """Evaluates raw predictions (mirrors ydf.evaluate_predictions).
    `weights` makes accuracy/loss/rmse/mae example-weighted (reference
    weighted evaluation, metric.cc)."""
    from ydf_amd.dataset.dataspec import Task

    ev = Evaluation(num_examples=len(labels))
    w = None if weights is None else np.asarray(weights, np.float64)
    ev.num_examples_weighted = float(w.sum()) if w is not None \
        else float(len(labels))
    if task == Task.CLASSIFICATION:
        if predictions.ndim == 1:
            pred_cls = (predictions >= 0.5).astype(np.int64)
            ev.auc = roc_auc(labels > 0.5, predictions)
            fpr, tpr, thr = roc_curve(labels > 0.5, predictions)
            ev.pr_auc = pr_auc(labels > 0.5, predictions)
            np_ = int((labels > 0.5).sum())
            ev.characteristics = [Characteristic(
                name="default", fpr=fpr, tpr=tpr, thresholds=thr,
                n_pos=np_, n_neg=len(labels) - np_, roc_auc=ev.auc,
                pr_auc=ev.pr_auc)]
        else:
            pred_cls = predictions.argmax(axis=1)
        correct = (labels.astype(np.int64) == pred_cls)
        if w is None:
            ev.accuracy = accuracy(labels.astype(np.int64), pred_cls)
            ev.loss = log_loss(labels, predictions)
        else:
            ev.accuracy = float((correct * w).sum() / w.sum())
            if predictions.ndim == 1:
                p = np.clip(predictions.astype(np.float64), 1e-12,
                            1 - 1e-12)
                y = labels.astype(np.float64)
                per = -(y * np.log(p) + (1 - y) * np.log(1 - p))
            else:
                per = -np.log(np.clip(
                    predictions[np.arange(len(labels)),
                                labels.astype(int)].astype(np.float64),
                    1e-12, 1.0))
            ev.loss = float((per * w).sum() / w.sum())
        ev.confusion = confusion_matrix(labels.astype(np.int64), pred_cls,
                                        n_classes)
        ev.accuracy_ci95 = accuracy_confidence_interval(
            ev.accuracy, len(labels))
        if ev.auc is not None and np.isfinite(ev.auc):
            n_pos = int((labels > 0.5).sum())
            ev.auc_ci95 = auc_confidence_interval(
                ev.auc, n_pos, len(labels) - n_pos)
    elif task == Task.ANOMALY_DETECTION:
        # labels: 1 = anomaly; predictions: anomaly score in [0, 1]
        ev.auc = roc_auc(labels > 0.5, predictions)
        fpr, tpr, thr = roc_curve(labels > 0.5, predictions)
        ev.pr_auc = pr_auc(labels > 0.5, predictions)
        np_ = int((labels > 0.5).sum())
        ev.characteristics = [Characteristic(
            name="default", fpr=fpr, tpr=tpr, thresholds=thr,
            n_pos=np_, n_neg=len(labels) - np_, roc_auc=ev.auc,
            pr_auc=ev.pr_auc)]
    elif task == Task.REGRESSION:
        if w is None:
            ev.rmse = rmse(labels, predictions)
            ev.mae = mae(labels, predictions)
        else:
            err = (labels - predictions).astype(np.float64)
            ev.rmse = float(np.sqrt((err ** 2 * w).sum() / w.sum()))
            ev.mae = float((np.abs(err) * w).sum() / w.sum())
        ev.loss = ev.rmse ** 2
        # bootstrap 95% CI on RMSE (reference
        # bootstrap_rmse_lower/upper_bounds_95p, 199 resamples)
        se = ((labels - predictions).astype(np.float64)) ** 2
        if len(se) > 1:
            rng = np.random.RandomState(1234)
            idx = rng.randint(0, len(se), size=(199, len(se)))
            boots = np.sqrt(se[idx].mean(axis=1))
            ev.rmse_ci95_bootstrap = (
                float(np.percentile(boots, 2.5)),
                float(np.percentile(boots, 97.5)))
    return ev
