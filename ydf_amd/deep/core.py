"""Deep tabular learners (capability analogue of PYDF's ydf/deep/:
generic_jax.py GenericJaxLearner / GenericJAXModel, mlp.py,
tabular_transformer.py) implemented natively in PyTorch-ROCm — the
models train and serve on MI355X through torch (rocBLAS/MIOpen) instead
of flax/JAX, with the same Learner/Model API as the tree learners.
"""
from __future__ import annotations

import dataclasses
import json
import math
import os
import time
from typing import Dict, List, Optional

import numpy as np
import torch
from torch import nn

from ydf_amd.dataset.dataset import _to_column_dict
from ydf_amd.dataset.dataspec import Task
from ydf_amd.metric.metric import Evaluation, evaluate_predictions
from ydf_amd.utils.log import info


@dataclasses.dataclass
class FeatureSpec:
    name: str
    kind: str                    # "num" | "cat"
    mean: float = 0.0
    std: float = 1.0
    vocab: Optional[List[str]] = None

    def to_json(self):
        return dataclasses.asdict(self)

    @staticmethod
    def from_json(d):
        return FeatureSpec(**d)


def _infer_specs(cols: Dict[str, np.ndarray], label: str,
                 features: Optional[List[str]] = None
                 ) -> List[FeatureSpec]:
    specs = []
    names = features if features is not None else \
        [k for k in cols if k != label]
    for name in names:
        arr = cols[name]
        if arr.dtype.kind in "fiub":
            x = np.asarray(arr, dtype=np.float64)
            x = x[np.isfinite(x)]
            mean = float(x.mean()) if len(x) else 0.0
            std = float(x.std()) if len(x) else 1.0
            specs.append(FeatureSpec(name, "num", mean,
                                     std if std > 1e-12 else 1.0))
        else:
            vals, counts = np.unique(arr.astype(str), return_counts=True)
            order = np.argsort(-counts, kind="stable")
            vocab = [str(vals[i]) for i in order][:2000]
            specs.append(FeatureSpec(name, "cat", vocab=vocab))
    return specs


def _encode(cols: Dict[str, np.ndarray], specs: List[FeatureSpec]):
    """-> (num [N, n_num] f32 z-scored, cat [N, n_cat] i64)."""
    n = len(next(iter(cols.values())))
    nums, cats = [], []
    for s in specs:
        if s.kind == "num":
            x = np.asarray(cols[s.name], dtype=np.float32).copy()
            bad = ~np.isfinite(x)
            x[bad] = s.mean
            nums.append((x - s.mean) / s.std)
        else:
            lut = {v: i + 1 for i, v in enumerate(s.vocab)}
            cats.append(np.fromiter(
                (lut.get(str(v), 0) for v in cols[s.name]),
                dtype=np.int64, count=n))
    num = np.stack(nums, 1) if nums else np.zeros((n, 0), np.float32)
    cat = np.stack(cats, 1) if cats else np.zeros((n, 0), np.int64)
    return num.astype(np.float32), cat


class _MLPNet(nn.Module):
    """mlp.py:76 MultiLayerPerceptronImpl, in torch."""

    def __init__(self, n_num: int, cat_vocab: List[int], n_out: int,
                 num_layers: int, layer_size: int, drop_out: float):
        super().__init__()
        self.embs = nn.ModuleList(
            [nn.Embedding(v + 1, min(16, max(2, int(v ** 0.5) + 1)))
             for v in cat_vocab])
        d_in = n_num + sum(e.embedding_dim for e in self.embs)
        layers: List[nn.Module] = []
        d = d_in
        for _ in range(max(1, num_layers - 1)):
            layers += [nn.Linear(d, layer_size), nn.ReLU(),
                       nn.Dropout(drop_out)]
            d = layer_size
        self.body = nn.Sequential(*layers)
        self.head = nn.Linear(d, n_out)

    def forward(self, num, cat):
        parts = [num] + [emb(cat[:, i])
                         for i, emb in enumerate(self.embs)]
        return self.head(self.body(torch.cat(parts, dim=1)))


class _TransformerNet(nn.Module):
    """tabular_transformer.py:83 TabularTransformerImpl, in torch: one
    token per feature + CLS token, standard encoder stack."""

    def __init__(self, n_num: int, cat_vocab: List[int], n_out: int,
                 num_layers: int, token_dim: int, num_heads: int,
                 drop_out: float):
        super().__init__()
        self.n_num = n_num
        self.num_proj = nn.Parameter(torch.randn(n_num, token_dim) * 0.02)
        self.num_bias = nn.Parameter(torch.zeros(n_num, token_dim))
        self.embs = nn.ModuleList(
            [nn.Embedding(v + 1, token_dim) for v in cat_vocab])
        self.cls = nn.Parameter(torch.zeros(1, 1, token_dim))
        layer = nn.TransformerEncoderLayer(
            d_model=token_dim, nhead=num_heads,
            dim_feedforward=4 * token_dim, dropout=drop_out,
            batch_first=True, norm_first=True)
        self.encoder = nn.TransformerEncoder(
            layer, num_layers, enable_nested_tensor=False)
        self.head = nn.Linear(token_dim, n_out)

    def forward(self, num, cat):
        toks = []
        if self.n_num:
            toks.append(num.unsqueeze(-1) * self.num_proj + self.num_bias)
        if len(self.embs):
            toks.append(torch.stack(
                [emb(cat[:, i]) for i, emb in enumerate(self.embs)],
                dim=1))
        x = torch.cat(toks, dim=1) if toks else num.new_zeros(
            (num.shape[0], 0, self.cls.shape[-1]))
        cls = self.cls.expand(x.shape[0], 1, -1)
        x = torch.cat([cls, x], dim=1)
        return self.head(self.encoder(x)[:, 0])


class DeepModel:
    """Serving wrapper (PYDF GenericJAXModel analogue): predict /
    evaluate / save / load over a torch module."""

    def __init__(self, net: nn.Module, specs: List[FeatureSpec],
                 task: Task, label: str,
                 label_classes: Optional[List[str]],
                 label_mean: float, label_std: float, model_type: str,
                 config: Dict):
        self.net = net
        self.specs = specs
        self._task = task
        self._label = label
        self.label_classes = label_classes
        self.label_mean = label_mean
        self.label_std = label_std
        self._model_type = model_type
        self.config = config
        self.training_logs: List[Dict] = []

    def name(self) -> str:
        return self._model_type

    def task(self) -> Task:
        return self._task

    def label(self) -> str:
        return self._label

    def input_feature_names(self) -> List[str]:
        return [s.name for s in self.specs]

    def predict(self, data, device=None, batch_size: int = 8192
                ) -> np.ndarray:
        dev = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        cols = _to_column_dict(data)
        num, cat = _encode(cols, self.specs)
        self.net.to(dev).eval()
        outs = []
        with torch.no_grad():
            for i in range(0, len(num), batch_size):
                nb = torch.from_numpy(num[i:i + batch_size]).to(dev)
                cb = torch.from_numpy(cat[i:i + batch_size]).to(dev)
                outs.append(self.net(nb, cb).float().cpu())
        raw = torch.cat(outs).numpy()
        if self._task == Task.CLASSIFICATION:
            if raw.shape[1] == 1:
                return 1.0 / (1.0 + np.exp(-raw[:, 0]))
            e = np.exp(raw - raw.max(1, keepdims=True))
            return e / e.sum(1, keepdims=True)
        return raw[:, 0] * self.label_std + self.label_mean

    def evaluate(self, data, device=None) -> Evaluation:
        cols = _to_column_dict(data)
        preds = self.predict(data, device=device)
        y = cols[self._label]
        if self._task == Task.CLASSIFICATION:
            lut = {v: i for i, v in enumerate(self.label_classes)}
            labels = np.fromiter((lut.get(str(v), 0) for v in y),
                                 dtype=np.float32, count=len(y))
            return evaluate_predictions(preds, labels, self._task,
                                        len(self.label_classes))
        return evaluate_predictions(
            preds, np.asarray(y, np.float32), self._task)

    def describe(self, output_format: str = "text") -> str:
        n_params = sum(p.numel() for p in self.net.parameters())
        return (f"{self._model_type} model\nfeatures: "
                f"{len(self.specs)}\nparameters: {n_params}\n"
                f"config: {self.config}")

    def save(self, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        header = {"model_type": self._model_type,
                  "task": self._task.name, "label": self._label,
                  "label_classes": self.label_classes,
                  "label_mean": self.label_mean,
                  "label_std": self.label_std,
                  "config": self.config,
                  "specs": [s.to_json() for s in self.specs]}
        with open(os.path.join(path, "header.json"), "w") as f:
            json.dump(header, f, indent=1)
        torch.save(self.net.state_dict(),
                   os.path.join(path, "weights.pt"))
        with open(os.path.join(path, "done"), "w") as f:
            f.write("")

    @staticmethod
    def load(path: str) -> "DeepModel":
        with open(os.path.join(path, "header.json")) as f:
            header = json.load(f)
        specs = [FeatureSpec.from_json(d) for d in header["specs"]]
        cfg = header["config"]
        task = Task[header["task"]]
        n_num = sum(1 for s in specs if s.kind == "num")
        cat_vocab = [len(s.vocab) for s in specs if s.kind == "cat"]
        n_out = cfg["n_out"]
        if header["model_type"] == "TABULAR_TRANSFORMER":
            net = _TransformerNet(n_num, cat_vocab, n_out,
                                  cfg["num_layers"], cfg["token_dim"],
                                  cfg["num_heads"], cfg["drop_out"])
        else:
            net = _MLPNet(n_num, cat_vocab, n_out, cfg["num_layers"],
                          cfg["layer_size"], cfg["drop_out"])
        net.load_state_dict(torch.load(
            os.path.join(path, "weights.pt"), map_location="cpu"))
        return DeepModel(net, specs, task, header["label"],
                         header.get("label_classes"),
                         header.get("label_mean", 0.0),
                         header.get("label_std", 1.0),
                         header["model_type"], cfg)


class _GenericDeepLearner:
    """Shared train loop (PYDF generic_jax.GenericJaxLearner:
    cosine-decay Adam, epoch-patience early stopping with parameter
    revert, max-duration)."""

    model_type = "MLP"

    def __init__(self, label: str, task: Task = Task.CLASSIFICATION,
                 features=None, batch_size: int = 512,
                 num_epochs: int = 1000, learning_rate: float = 0.01,
                 learning_rate_policy: str = "cosine_decay",
                 num_steps: Optional[int] = None,
                 maximum_training_duration_seconds: float = -1.0,
                 early_stopping_epoch_patience: Optional[int] = 10,
                 early_stopping_revert_params: bool = True,
                 validation_ratio: float = 0.1,
                 random_seed: int = 1234, device=None, **net_kwargs):
        self.label = label
        self._task = task
        self.features = features
        self.batch_size = batch_size
        self.num_epochs = num_epochs
        self.learning_rate = learning_rate
        self.learning_rate_policy = learning_rate_policy
        self.num_steps = num_steps
        self.max_duration = maximum_training_duration_seconds
        self.patience = early_stopping_epoch_patience
        self.revert = early_stopping_revert_params
        self.validation_ratio = validation_ratio
        self.random_seed = random_seed
        self.device = device
        self.net_kwargs = net_kwargs

    def _make_net(self, n_num, cat_vocab, n_out) -> nn.Module:
        raise NotImplementedError

    def train(self, data, valid=None, verbose=None) -> DeepModel:
        dev = torch.device(self.device) if self.device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        torch.manual_seed(self.random_seed)
        cols = _to_column_dict(data)
        specs = _infer_specs(cols, self.label, self.features)
        num, cat = _encode(cols, specs)
        y_raw = cols[self.label]
        label_classes = None
        label_mean, label_std = 0.0, 1.0
        if self._task == Task.CLASSIFICATION:
            vals, counts = np.unique(np.asarray(y_raw).astype(str),
                                     return_counts=True)
            order = np.argsort(-counts, kind="stable")
            label_classes = [str(vals[i]) for i in order]
            lut = {v: i for i, v in enumerate(label_classes)}
            y = np.fromiter((lut[str(v)] for v in y_raw),
                            dtype=np.int64, count=len(y_raw))
            n_out = 1 if len(label_classes) == 2 else len(label_classes)
        elif self._task == Task.REGRESSION:
            yf = np.asarray(y_raw, np.float64)
            label_mean = float(yf.mean())
            label_std = float(yf.std()) or 1.0
            y = ((yf - label_mean) / label_std).astype(np.float32)
            n_out = 1
        else:
            raise NotImplementedError(
                f"deep learners support classification/regression, "
                f"got {self._task}")

        N = len(y)
        rng = np.random.RandomState(self.random_seed)
        n_valid = int(N * self.validation_ratio) \
            if self.patience is not None and N > 50 else 0
        perm = rng.permutation(N)
        vi, ti = perm[:n_valid], perm[n_valid:]
        n_num = num.shape[1]
        cat_vocab = [len(s.vocab) for s in specs if s.kind == "cat"]
        net = self._make_net(n_num, cat_vocab, n_out).to(dev)
        opt = torch.optim.Adam(net.parameters(), lr=self.learning_rate)
        steps_per_epoch = max(1, (len(ti) + self.batch_size - 1)
                              // self.batch_size)
        total_steps = self.num_steps or self.num_epochs * steps_per_epoch
        sched = torch.optim.lr_scheduler.CosineAnnealingLR(
            opt, T_max=total_steps) \
            if self.learning_rate_policy == "cosine_decay" else None
        if self._task == Task.CLASSIFICATION:
            loss_fn = nn.BCEWithLogitsLoss() if n_out == 1 \
                else nn.CrossEntropyLoss()
            yt = torch.from_numpy(y.astype(np.float32) if n_out == 1
                                  else y).to(dev)
        else:
            loss_fn = nn.MSELoss()
            yt = torch.from_numpy(y).to(dev)
        num_t = torch.from_numpy(num).to(dev)
        cat_t = torch.from_numpy(cat).to(dev)
        ti_t = torch.from_numpy(ti).to(dev)
        vi_t = torch.from_numpy(vi).to(dev)

        def batch_loss(idx):
            out = net(num_t[idx], cat_t[idx])
            tgt = yt[idx]
            if n_out == 1:
                return loss_fn(out[:, 0], tgt)
            return loss_fn(out, tgt)

        best_val = math.inf
        best_state = None
        bad_epochs = 0
        step = 0
        t0 = time.monotonic()
        logs = []
        stop = False
        for epoch in range(self.num_epochs):
            if stop:
                break
            net.train()
            order = ti_t[torch.randperm(len(ti_t), device=dev)]
            for i in range(0, len(order), self.batch_size):
                idx = order[i:i + self.batch_size]
                opt.zero_grad(set_to_none=True)
                loss = batch_loss(idx)
                loss.backward()
                opt.step()
                if sched is not None:
                    sched.step()
                step += 1
                if self.num_steps and step >= self.num_steps:
                    stop = True
                    break
            if self.max_duration > 0 and \
                    time.monotonic() - t0 > self.max_duration:
                info(f"max duration reached at epoch {epoch}")
                stop = True
            if n_valid:
                net.eval()
                with torch.no_grad():
                    vloss = float(batch_loss(vi_t).item())
                logs.append({"epoch": epoch + 1, "valid_loss": vloss})
                if vloss < best_val - 1e-6:
                    best_val = vloss
                    bad_epochs = 0
                    if self.revert:
                        best_state = {k: v.detach().clone()
                                      for k, v in
                                      net.state_dict().items()}
                else:
                    bad_epochs += 1
                    if self.patience is not None and \
                            bad_epochs >= self.patience:
                        info(f"early stop at epoch {epoch + 1}")
                        break
        if best_state is not None:
            net.load_state_dict(best_state)
        cfg = dict(self.net_kwargs)
        cfg["n_out"] = n_out
        model = DeepModel(net.cpu(), specs, self._task, self.label,
                          label_classes, label_mean, label_std,
                          self.model_type, cfg)
        model.training_logs = logs
        return model


class MultiLayerPerceptronLearner(_GenericDeepLearner):
    """mlp.py:120 MultiLayerPerceptronLearner (torch)."""

    model_type = "MLP"

    def __init__(self, label: str, task: Task = Task.CLASSIFICATION,
                 num_layers: int = 8, layer_size: int = 200,
                 drop_out: float = 0.05, **kwargs):
        super().__init__(label=label, task=task, num_layers=num_layers,
                         layer_size=layer_size, drop_out=drop_out,
                         **kwargs)

    def _make_net(self, n_num, cat_vocab, n_out):
        k = self.net_kwargs
        return _MLPNet(n_num, cat_vocab, n_out, k["num_layers"],
                       k["layer_size"], k["drop_out"])


class TabularTransformerLearner(_GenericDeepLearner):
    """tabular_transformer.py:292 TabularTransformerLearner (torch)."""

    model_type = "TABULAR_TRANSFORMER"

    def __init__(self, label: str, task: Task = Task.CLASSIFICATION,
                 num_layers: int = 3, token_dim: int = 50,
                 num_heads: int = 4, drop_out: float = 0.05,
                 batch_size: int = 256, learning_rate: float = 0.001,
                 **kwargs):
        # token_dim must be divisible by num_heads for the attention
        token_dim = ((token_dim + num_heads - 1) // num_heads) * num_heads
        super().__init__(label=label, task=task, batch_size=batch_size,
                         learning_rate=learning_rate,
                         num_layers=num_layers, token_dim=token_dim,
                         num_heads=num_heads, drop_out=drop_out, **kwargs)

    def _make_net(self, n_num, cat_vocab, n_out):
        k = self.net_kwargs
        return _TransformerNet(n_num, cat_vocab, n_out, k["num_layers"],
                               k["token_dim"], k["num_heads"],
                               k["drop_out"])
