"""NUMERICAL_VECTOR_SEQUENCE columns: ragged vector-sequence cells.

Reference analogue: data_spec.proto:73-81 NumericalVectorSequence
columns and the reference's only GPU code
(learner/decision_tree/gpu.cu.cc:46-136 KernelComputeMaxDotProduct /
KernelComputeNegMinSquareDistance; conditions in
decision_tree.proto:133-161 CloserThan / ProjectedMoreThan).

MI355X design: anchors are sampled once per dataspec (seeded) from
observed vectors; each (anchor, kind) becomes a VIRTUAL NUMERICAL
column holding
  kind "dot":  max_k <vec_k, anchor>        (ProjectedMoreThan >= thr)
  kind "dist": -min_k |vec_k - anchor|^2    (CloserThan |.|^2 <= thr2,
              negated so the condition keeps the ">= threshold" shape)
so the binned histogram kernels, the serving kernels and the exact
oracle all consume vector-sequence features with no special cases —
only the projection itself is a dedicated kernel (ops.vecseq_project,
LDS-staged anchors, wave64 example tiles). Empty sequences project to
-3e38: every "exists" condition evaluates false, matching the
reference's semantics. Deviation (documented): the reference samples
anchors per candidate split; sampling per dataspec trades per-node
adaptivity for fully-reusable projection columns.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np

EMPTY_PROJ = np.float32(-3.0e38)


def is_vecseq_cell(cell) -> bool:
    if isinstance(cell, np.ndarray):
        return cell.ndim == 2
    if isinstance(cell, (list, tuple)) and cell:
        return isinstance(cell[0], (list, tuple, np.ndarray))
    return False


def extract_ragged(arr) -> Tuple[np.ndarray, np.ndarray, int]:
    """Object array of [n_k, dim] cells -> (values [K, dim] f32,
    offs i64 [N+1], dim). Empty/None cells contribute zero vectors."""
    mats = []
    lens = np.zeros(len(arr) + 1, dtype=np.int64)
    dim = 0
    for i, cell in enumerate(arr):
        if cell is None:
            continue
        m = np.asarray(cell, dtype=np.float32)
        if m.size == 0:
            continue
        if m.ndim == 1:
            m = m.reshape(1, -1)
        if dim == 0:
            dim = m.shape[1]
        elif m.shape[1] != dim:
            raise ValueError(
                f"inconsistent vector dim {m.shape[1]} vs {dim}")
        mats.append(m)
        lens[i + 1] = m.shape[0]
    offs = np.cumsum(lens)
    values = np.concatenate(mats, axis=0) if mats else \
        np.zeros((0, max(dim, 1)), dtype=np.float32)
    return np.ascontiguousarray(values, dtype=np.float32), offs, \
        max(dim, 1)


def sample_anchors(values: np.ndarray, n_anchors: int,
                   seed: int) -> np.ndarray:
    """Anchors = randomly selected observed vectors (reference
    num_random_selected_anchors sampling)."""
    rng = np.random.RandomState(seed)
    if len(values) == 0:
        return np.zeros((0, values.shape[1]), dtype=np.float32)
    idx = rng.randint(0, len(values), size=n_anchors)
    return np.ascontiguousarray(values[idx], dtype=np.float32)


def project_numpy(values: np.ndarray, offs: np.ndarray,
                  anchors: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
    """(maxdot [A,N], negminsq [A,N]) over each example's vector run."""
    N = len(offs) - 1
    A = len(anchors)
    maxdot = np.full((A, N), EMPTY_PROJ, dtype=np.float32)
    negminsq = np.full((A, N), EMPTY_PROJ, dtype=np.float32)
    if len(values) == 0 or A == 0:
        return maxdot, negminsq
    dots = values @ anchors.T                       # [K, A]
    vsq = (values * values).sum(axis=1)             # [K]
    asq = (anchors * anchors).sum(axis=1)           # [A]
    # -|v-a|^2 = 2 v.a - |v|^2 - |a|^2
    negs = 2.0 * dots - vsq[:, None] - asq[None, :]  # [K, A]
    nz = np.nonzero(offs[1:] > offs[:-1])[0]
    starts = offs[nz]
    md = np.maximum.reduceat(dots, starts, axis=0)
    ns = np.maximum.reduceat(negs, starts, axis=0)
    maxdot[:, nz] = md.T.astype(np.float32)
    negminsq[:, nz] = ns.T.astype(np.float32)
    return maxdot, negminsq


def virtual_specs(name: str, values: np.ndarray, offs: np.ndarray,
                  dim: int, n_anchors: int, seed: int,
                  numerical_boundaries_fn) -> List:
    """Builds the virtual projection ColumnSpecs (+ the parent marker
    spec is the caller's job)."""
    from ydf_amd.dataset.dataspec import ColumnSpec, Semantic

    anchors = sample_anchors(values, n_anchors, seed)
    maxdot, negminsq = project_numpy(values, offs, anchors)
    specs = []
    for j in range(len(anchors)):
        for kind, proj in (("dot", maxdot[j]), ("dist", negminsq[j])):
            live = proj[proj > EMPTY_PROJ / 2]
            specs.append(ColumnSpec(
                name=f"{name}.{kind}{j}",
                semantic=Semantic.NUMERICAL,
                mean=float(live.mean()) if live.size else 0.0,
                min_value=float(live.min()) if live.size else 0.0,
                max_value=float(live.max()) if live.size else 0.0,
                boundaries=numerical_boundaries_fn(proj),
                vecseq_source=name, vecseq_kind=kind,
                vecseq_anchor=anchors[j].copy()))
    return specs


def fill_vecseq_columns(X: np.ndarray, specs, cols: Dict[str, np.ndarray],
                        device=None) -> None:
    """Fills rows of X for every virtual vecseq spec, grouped by source
    column (one ragged extraction + one projection pass per source).
    On a CUDA device the projection runs through the HIP kernel."""
    by_src: Dict[str, List[Tuple[int, object]]] = {}
    for i, spec in enumerate(specs):
        if spec.vecseq_source is not None:
            by_src.setdefault(spec.vecseq_source, []).append((i, spec))
    for src, entries in by_src.items():
        if src not in cols:
            raise ValueError(f"missing input feature {src!r}")
        values, offs, dim = extract_ragged(cols[src])
        anchors = np.stack([s.vecseq_anchor for _, s in entries])
        # one projection per unique anchor row; entries alternate
        # dot/dist per anchor but we just compute both for all
        uniq, inv = np.unique(anchors, axis=0, return_inverse=True)
        if device is not None and getattr(device, "type", "") == "cuda":
            from ydf_amd import ops
            import torch

            vt = torch.from_numpy(values).to(device)
            ot = torch.from_numpy(offs).to(device)
            at = torch.from_numpy(
                np.ascontiguousarray(uniq, dtype=np.float32)).to(device)
            md_t, ns_t = ops.vecseq_project(vt, ot, at)
            maxdot, negminsq = md_t.cpu().numpy(), ns_t.cpu().numpy()
        else:
            maxdot, negminsq = project_numpy(
                values, offs, np.ascontiguousarray(uniq,
                                                   dtype=np.float32))
        for (row, spec), a_i in zip(entries, inv):
            X[row] = maxdot[a_i] if spec.vecseq_kind == "dot" \
                else negminsq[a_i]
