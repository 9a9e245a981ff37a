"""Cox proportional-hazards loss for survival analysis (capability
analogue of the reference loss_imp_cox.cc, task SURVIVAL_ANALYSIS with
label = event/censoring time, label_event_observed = indicator, optional
entry age for left truncation).

Breslow formulation over margins m (log relative hazard):
  L = -(1/n) sum_{i: event} [ m_i - log sum_{j in R_i} exp(m_j) ]
with risk set R_i = {j : entry_j < T_i <= T_j}. Gradients/hessians are
computed in O(N log N) with sorted suffix/prefix sums (vectorized torch,
runs on the training device).
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch


class CoxData:
    """Precomputed ordering for one dataset split."""

    def __init__(self, times: np.ndarray, events: np.ndarray, device,
                 entry_ages: Optional[np.ndarray] = None):
        self.N = len(times)
        t = np.asarray(times, dtype=np.float64)
        e = np.asarray(events, dtype=bool)
        # descending time order: suffix sums over exp(m) become prefix
        order = np.argsort(-t, kind="stable")
        self.order = torch.from_numpy(order).to(device)
        self.t_sorted = torch.from_numpy(t[order]).to(device)
        self.e_sorted = torch.from_numpy(e[order]).to(device)
        self.device = device
        self.entry = None
        if entry_ages is not None:
            self.entry = torch.from_numpy(
                np.asarray(entry_ages, dtype=np.float64)[order]).to(device)

    def _risk_denoms(self, m_sorted: torch.Tensor) -> torch.Tensor:
        """S_i = sum_{j: T_j >= T_i} exp(m_j) for each sorted position
        (ties share the suffix sum). Left truncation subtracts members
        whose entry age >= T_i (not yet at risk)."""
        ex = torch.exp((m_sorted - m_sorted.max()).clamp(min=-60.0))
        csum = torch.cumsum(ex, dim=0)
        # positions with equal time must share the FULL tied sum: take
        # the last index of each tied run
        t = self.t_sorted
        n = len(t)
        last_of_run = torch.ones(n, dtype=torch.bool, device=t.device)
        last_of_run[:-1] = t[:-1] != t[1:]
        run_id = torch.cumsum(last_of_run.long(), dim=0) \
            - last_of_run.long()
        run_last_idx = torch.nonzero(last_of_run).view(-1)
        S = csum[run_last_idx][run_id]
        if self.entry is not None:
            # remove exp(m_j) of subjects with entry_j >= T_i: sort by
            # entry descending and accumulate (approximation: exact for
            # entry < T constraints evaluated pairwise via searchsorted)
            ent_sorted, ent_order = torch.sort(self.entry, descending=True)
            ex_by_entry = ex[ent_order]
            centry = torch.cumsum(ex_by_entry, dim=0)
            # for each i: number of subjects with entry >= T_i
            k = torch.searchsorted(-ent_sorted.contiguous(),
                                   -t.contiguous(), right=True)
            sub = torch.where(k > 0, centry[(k - 1).clamp(min=0)],
                              torch.zeros_like(S))
            S = (S - sub).clamp(min=1e-30)
        return S, ex

    def grad_hess(self, m: torch.Tensor) -> Tuple[torch.Tensor,
                                                  torch.Tensor]:
        ms = m.double()[self.order]
        S, ex = self._risk_denoms(ms)
        invS = torch.where(self.e_sorted, 1.0 / S, torch.zeros_like(S))
        inv2 = torch.where(self.e_sorted, 1.0 / (S * S),
                           torch.zeros_like(S))
        # A_k = sum over events i with T_i <= T_k of 1/S_i  (events appear
        # at positions >= k in descending order -> suffix sums), but ties
        # must include same-time events: use run boundaries
        t = self.t_sorted
        n = len(t)
        first_of_run = torch.ones(n, dtype=torch.bool, device=t.device)
        first_of_run[1:] = t[1:] != t[:-1]
        run_id = torch.cumsum(first_of_run.long(), dim=0) - 1
        sufA = torch.flip(torch.cumsum(torch.flip(invS, [0]), 0), [0])
        sufB = torch.flip(torch.cumsum(torch.flip(inv2, [0]), 0), [0])
        run_first_idx = torch.nonzero(first_of_run).view(-1)
        A = sufA[run_first_idx][run_id]
        B = sufB[run_first_idx][run_id]
        w = 1.0 / self.N
        g_sorted = w * (ex * A - self.e_sorted.double())
        h_sorted = w * (ex * A - ex * ex * B)
        g = torch.empty_like(g_sorted)
        h = torch.empty_like(h_sorted)
        g[self.order] = g_sorted
        h[self.order] = h_sorted
        return (g.float(),
                h.float().clamp(1e-8, 16.0))

    def loss(self, m: torch.Tensor) -> float:
        ms = m.double()[self.order]
        shift = ms.max()
        S, _ = self._risk_denoms(ms)
        ll = torch.where(self.e_sorted,
                         ms - shift - torch.log(S.clamp(min=1e-30)),
                         torch.zeros_like(ms))
        return float((-ll.sum() / self.N).item())
