"""Metric layer exports (mirrors the ydf.metric namespace)."""
from ydf_amd.metric.metric import (Characteristic, ConfusionMatrix,  # noqa
                                   Evaluation, accuracy,
                                   bootstrap_confidence_intervals,
                                   confusion_matrix, log_loss, mae,
                                   mean_average_precision, mrr, ndcg,
                                   pr_auc, rmse, roc_auc, roc_curve)
