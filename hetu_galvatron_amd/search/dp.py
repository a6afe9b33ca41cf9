"""Knapsack-style layer-wise DP over (layer, memory, strategy).

Reference: galvatron/core/search_engine/dynamic_programming.py:12-115
(DPAlg.fit) and csrc/dp_core.cpp:24-120 (C++ core).  Same recurrence:

    f[v][s] = min_si f[v - v_data[i][s]][si] + inter[i][si][s] + intra[i][s]

Differences by design: memory is bucketed in `mem_unit_mb` units (MI355X
budgets are 100s of GB — MB-granular tables would be ~1 GB of marks), and
the C++ core (_galvatron_dp_core, csrc_cpu/dp_core.cpp) fills f/mark
buffers handed in from numpy while the back-trace stays in Python.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np

try:
    import torch  # noqa: F401  (the extension links torch's C++ libs)
    from .. import _galvatron_dp_core as _core
except ImportError:  # pure-python fallback mirrors the C++ core
    _core = None


def solve_layer_dp(v_data: np.ndarray, intra: np.ndarray, inter: np.ndarray,
                   max_mem_units: int, use_cpp: bool = True
                   ) -> Tuple[np.ndarray, np.ndarray]:
    """Fill the DP tables.

    v_data [L,S] int32 (memory units), intra [L,S] ms, inter [L,S,S] ms
    (inter[i][si][s]: layer i-1 ran si, layer i runs s).
    Returns f [M+1, S] float64 and mark [L, M+1, S] int16.
    """
    L, S = v_data.shape
    M = max_mem_units + 1
    f = np.zeros((M, S), dtype=np.float64)
    mark = np.full((L, M, S), -1, dtype=np.int16)
    if use_cpp and _core is not None:
        _core.dynamic_programming_core(
            int(L), int(M), int(S),
            np.ascontiguousarray(v_data, dtype=np.int32),
            np.ascontiguousarray(intra, dtype=np.float64),
            np.ascontiguousarray(inter, dtype=np.float64),
            f, mark)
        return f, mark
    # python fallback (vectorized over memory levels)
    INF = np.inf
    for i in range(L):
        nf = np.full((M, S), INF)
        for s in range(S):
            vd = int(v_data[i, s])
            if vd >= M:
                continue
            # candidate over predecessor strategy si at budget v - vd
            base = f[: M - vd, :] + inter[i, :, s][None, :]  # [M-vd, S]
            best_si = np.argmin(base, axis=1)
            best = base[np.arange(M - vd), best_si] + intra[i, s]
            nf[vd:, s] = best
            mark[i, vd:, s] = best_si.astype(np.int16)
        f = nf
    return f, mark


def backtrace(v_data: np.ndarray, mark: np.ndarray, f: np.ndarray,
              budget_units: int) -> Tuple[float, Optional[List[int]], int]:
    """Best strategy path within `budget_units`. Returns (cost, path, leftover)."""
    L, S = v_data.shape
    v = min(budget_units, f.shape[0] - 1)
    if v < 0:
        return np.inf, None, -1
    s = int(np.argmin(f[v, :]))
    cost = float(f[v, s])
    if not np.isfinite(cost):
        return np.inf, None, -1
    path = [-1] * L
    path[-1] = s
    for i in range(L - 1, 0, -1):
        si = int(mark[i, v, s])
        v -= int(v_data[i, s])
        s = si
        path[i - 1] = s
    v -= int(v_data[0, s])
    return cost, path, v
