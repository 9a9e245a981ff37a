"""LambdaMART-NDCG ranking gradients (capability analogue of the
reference's LAMBDA_MART_NDCG loss,
learner/gradient_boosted_trees/loss/loss_imp_ndcg.*).

Queries are padded into a dense [n_groups, max_group] layout once; per
boosting iteration the pairwise lambda gradients are computed with batched
torch ops (runs on the training device — GPU tensors on MI355X)."""
from __future__ import annotations

from typing import Tuple

import numpy as np
import torch


class RankingLambdas:
    """Precomputes the padded group layout and per-query ideal DCG."""

    def __init__(self, group_ids: np.ndarray, relevances: np.ndarray,
                 device, truncation: int = 5, sigma: float = 1.0):
        order = np.argsort(group_ids, kind="stable")
        assert (order == np.arange(len(order))).all(), \
            "rows must be pre-sorted by group (learner sorts at ingestion)"
        uniq, starts = np.unique(group_ids, return_index=True)
        starts = np.sort(starts)
        ends = np.append(starts[1:], len(group_ids))
        sizes = ends - starts
        self.Q = len(starts)
        self.M = int(sizes.max())
        self.truncation = truncation
        self.sigma = sigma
        idx = np.full((self.Q, self.M), -1, dtype=np.int64)
        for q, (s, e) in enumerate(zip(starts, ends)):
            idx[q, : e - s] = np.arange(s, e)
        self.idx = torch.from_numpy(idx).to(device)
        self.valid = self.idx >= 0
        self.safe_idx = self.idx.clamp(min=0)
        rel = torch.from_numpy(
            np.ascontiguousarray(relevances.astype(np.float32))).to(device)
        self.rel = torch.where(
            self.valid, rel[self.safe_idx],
            torch.full((), -1.0, device=device))
        self.gains = torch.where(self.valid, 2.0 ** self.rel - 1.0,
                                 torch.zeros((), device=device))
        # ideal DCG at truncation
        sorted_gains, _ = torch.sort(self.gains, dim=1, descending=True)
        disc = self._discounts(device)
        self.idcg = (sorted_gains[:, : self.M] * disc).sum(dim=1).clamp(1e-9)
        self.sizes = self.valid.sum(dim=1)
        self.N = len(group_ids)
        self.device = device
        # positive examples (relevance > 0) — used by SelGB sampling
        pm = torch.zeros(self.N, dtype=torch.bool, device=device)
        pm[self.safe_idx[self.valid & (self.rel > 0)]] = True
        self.positive_mask = pm

    def _discounts(self, device):
        r = torch.arange(self.M, device=device, dtype=torch.float32)
        d = 1.0 / torch.log2(r + 2.0)
        d = torch.where(r < self.truncation, d, torch.zeros_like(d))
        return d

    def lambdas(self, scores: torch.Tensor) -> Tuple[torch.Tensor,
                                                     torch.Tensor]:
        """scores [N] -> (g, h) [N] LambdaMART gradients/hessians."""
        S = torch.where(self.valid, scores[self.safe_idx],
                        torch.full((), -1e30, device=self.device))
        # rank of each item within its query (0-based, by descending score)
        order = torch.argsort(S, dim=1, descending=True, stable=True)
        ranks = torch.empty_like(order)
        ar = torch.arange(self.M, device=self.device).expand_as(order)
        ranks.scatter_(1, order, ar)
        disc = self._discounts(self.device)[ranks.clamp(max=self.M - 1)]
        disc = torch.where(self.valid, disc, torch.zeros_like(disc))
        # pairwise |delta NDCG| for swapping i and j
        gd = self.gains / self.idcg.unsqueeze(1)        # normalized gains
        dg = gd.unsqueeze(2) - gd.unsqueeze(1)          # [Q,M,M]
        dd = disc.unsqueeze(2) - disc.unsqueeze(1)
        delta = (dg * dd).abs()
        rel_diff = self.rel.unsqueeze(2) - self.rel.unsqueeze(1)
        pair = (rel_diff > 0)                            # i more relevant
        pair &= self.valid.unsqueeze(2) & self.valid.unsqueeze(1)
        sdiff = S.unsqueeze(2) - S.unsqueeze(1)
        rho = torch.sigmoid(-self.sigma * sdiff)         # P(j beats i)
        lam = torch.where(pair, self.sigma * rho * delta,
                          torch.zeros_like(rho))
        hess = torch.where(pair,
                           self.sigma * self.sigma * rho * (1 - rho) * delta,
                           torch.zeros_like(rho))
        # item i gets -lam for every pair (i better), +lam when it is j
        g_mat = -lam.sum(dim=2) + lam.sum(dim=1)
        h_mat = hess.sum(dim=2) + hess.sum(dim=1)
        g = torch.zeros(self.N, dtype=torch.float32, device=self.device)
        h = torch.zeros(self.N, dtype=torch.float32, device=self.device)
        flat_idx = self.safe_idx[self.valid]
        g[flat_idx] = g_mat[self.valid]
        h[flat_idx] = h_mat[self.valid].clamp(1e-6, 16.0)
        return g, h

    def xe_ndcg(self, scores: torch.Tensor, seed: int):
        """Cross-entropy NDCG gradients (reference
        loss_imp_cross_entropy_ndcg.cc:104-180): softmax the per-group
        scores, target distribution (2^rel - gamma)/sum with gamma ~
        U(0,1) resampled per iteration, third-order Newton correction."""
        S = torch.where(self.valid, scores[self.safe_idx],
                        torch.full((), -1e30, device=self.device))
        p = torch.softmax(S, dim=1).clamp(1e-5, 0.99999)
        p = torch.where(self.valid, p, torch.zeros_like(p))
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed % (1 << 62))
        gamma = torch.rand((self.Q, self.M), generator=gen,
                           device=self.device)
        y = torch.where(self.valid, 2.0 ** self.rel - gamma,
                        torch.zeros_like(gamma))
        denom = y.sum(dim=1, keepdim=True)
        ok = ((denom != 0) & (self.sizes.unsqueeze(1) > 1)).float()
        denom = torch.where(denom == 0, torch.ones_like(denom), denom)
        t1 = -y / denom + p
        g_mat = -t1
        one_m_p = (1.0 - p).clamp(min=1e-5)
        l1 = t1 / one_m_p
        s1 = l1.sum(dim=1, keepdim=True)
        t2 = p * (s1 - l1)
        g_mat = g_mat - t2
        l2 = t2 / one_m_p
        s2 = l2.sum(dim=1, keepdim=True)
        g_mat = g_mat - p * (s2 - l2)
        h_mat = p * (1.0 - p)
        g_mat = g_mat * ok
        h_mat = h_mat * ok
        # reference fits trees to -gradient; our convention is g = dL/dm,
        # and the reference's gradient_data is the DESCENT direction
        g = torch.zeros(self.N, dtype=torch.float32, device=self.device)
        h = torch.zeros(self.N, dtype=torch.float32, device=self.device)
        flat_idx = self.safe_idx[self.valid]
        g[flat_idx] = -g_mat[self.valid]
        h[flat_idx] = h_mat[self.valid].clamp(1e-6, 16.0)
        return g, h

    def ndcg(self, scores: torch.Tensor) -> float:
        S = torch.where(self.valid, scores[self.safe_idx],
                        torch.full((), -1e30, device=self.device))
        order = torch.argsort(S, dim=1, descending=True, stable=True)
        gains_sorted = torch.gather(self.gains, 1, order)
        disc = self._discounts(self.device)
        dcg = (gains_sorted * disc).sum(dim=1)
        return float((dcg / self.idcg).mean().item())
