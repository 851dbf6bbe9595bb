"""Distributed greedy maximum-weight independent set.

Rebuild of ``util.py:12-51`` (``local_greedy_search`` — the repo's
Backpressure-lineage scheduler primitive; dead code in the reference but
part of the component inventory).  Vectorised: each round admits every
vertex whose weight beats all remaining neighbours (ties to the lower id),
then removes winners and their neighbourhoods.
"""

from __future__ import annotations

import numpy as np


def local_greedy_search(adj, wts):
    """adj: (N,N) dense or scipy sparse 0/1 adjacency; wts: (N,) weights.
    Returns (set of vertex ids, total weight)."""
    a = np.asarray(adj.todense() if hasattr(adj, "todense") else adj) != 0
    w = np.asarray(wts, dtype=np.float64).flatten()
    n = len(w)
    remain = np.ones(n, dtype=bool)
    mwis = set()
    while remain.any():
        wr = np.where(remain, w, -np.inf)
        # neighbour max among remaining (and the lowest id achieving it)
        nb = a & remain[None, :] & remain[:, None]
        nb_max = np.where(nb, wr[None, :], -np.inf).max(axis=1)
        ids = np.arange(n)
        better = remain & (wr > nb_max)
        # ties: vertex wins if its id is below every remaining neighbour
        # that attains the same weight
        tie = remain & (wr == nb_max) & np.isfinite(nb_max)
        for v in ids[tie]:
            nbs = ids[nb[v] & (wr[ids] == wr[v])]
            if len(nbs) == 0 or v < nbs.min():
                better[v] = True
        winners = ids[better]
        if len(winners) == 0:
            winners = ids[remain & (nb_max == -np.inf)]
            if len(winners) == 0:
                break
        for v in winners:
            if remain[v]:
                mwis.add(int(v))
                remain[v] = False
                remain[a[v]] = False
    return mwis, float(w[list(mwis)].sum())
