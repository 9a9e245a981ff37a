"""Survival metrics: Harrell's concordance index (reference reports Cox
loss; the C-index is the standard ranking quality for survival models)."""
from __future__ import annotations

import numpy as np


def concordance_index(times: np.ndarray, events: np.ndarray,
                      scores: np.ndarray) -> float:
    """Fraction of comparable pairs ordered correctly by risk score
    (higher score = higher hazard = earlier event); score ties 0.5."""
    t = np.asarray(times, np.float64)
    e = np.asarray(events, bool)
    s = np.asarray(scores, np.float64)
    n = len(t)
    # dense score ranks
    uniq, ranks = np.unique(s, return_inverse=True)
    m = len(uniq)
    tree = np.zeros(m + 1)      # Fenwick over score ranks (counts)

    def update(i):
        i += 1
        while i <= m:
            tree[i] += 1
            i += i & (-i)

    def query(i):  # count of inserted with rank < i
        tot = 0.0
        while i > 0:
            tot += tree[i]
            i -= i & (-i)
        return tot

    order = np.argsort(-t, kind="stable")
    conc = 0.0
    comp = 0
    inserted = 0
    i = 0
    while i < n:
        j = i
        while j < n and t[order[j]] == t[order[i]]:
            j += 1
        # query events of this tied-time group against STRICTLY later
        # times (already inserted)
        for k in range(i, j):
            idx = order[k]
            if not e[idx] or inserted == 0:
                continue
            r = ranks[idx]
            lower = query(r)            # later-time subjects, lower score
            eq = query(r + 1) - lower   # equal score
            # event (earlier time) should have the HIGHER hazard score
            conc += lower + 0.5 * eq
            comp += inserted
        for k in range(i, j):
            update(ranks[order[k]])
            inserted += 1
        i = j
    return conc / comp if comp else float("nan")
