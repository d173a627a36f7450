"""ILP address packing (MODeL-style) — optimal-peak alternative to the
min-skyline heuristic.

Capability parity with reference ``easydist/torch/schedule/
ilp_memory_scheduler.py`` (25+), re-based on scipy's HiGHS MILP (the
same solver the autoflow strategy MILP uses; the reference needed the
``mip`` package). Disjunctive non-overlap: for every lifetime-overlapping
pair either a sits below b or b below a (big-M with one binary per
pair); objective = peak. Buffer counts beyond ``max_ilp_buffers`` fall
back to the skyline packer (ILP is O(n²) binaries).
"""
from __future__ import annotations

import logging
from typing import Dict, List, Tuple

import numpy as np

from .efficient_memory_scheduler import ALIGN, _align, pack_buffers
from .lifetime import Buffer

logger = logging.getLogger(__name__)


def ilp_pack_buffers(buffers: List[Buffer], max_ilp_buffers: int = 60,
                     time_limit: float = 20.0
                     ) -> Tuple[Dict[Tuple[str, int], int], int]:
    if len(buffers) > max_ilp_buffers:
        logger.info("ilp_pack: %d buffers > cap %d, using skyline",
                    len(buffers), max_ilp_buffers)
        return pack_buffers(buffers)
    try:
        from scipy.optimize import LinearConstraint, milp
    except ImportError:
        return pack_buffers(buffers)

    n = len(buffers)
    sizes = [_align(b.size) for b in buffers]
    pairs = [(i, j) for i in range(n) for j in range(i + 1, n)
             if not (buffers[i].end < buffers[j].start
                     or buffers[j].end < buffers[i].start)]
    bigM = sum(sizes)
    # variables: o_0..o_{n-1}, P, y_p (one per pair)
    nv = n + 1 + len(pairs)
    c = np.zeros(nv)
    c[n] = 1.0                      # minimize peak

    lb: List[float] = []
    ub: List[float] = []
    A: List[np.ndarray] = []
    for i in range(n):              # o_i + s_i <= P
        r = np.zeros(nv)
        r[i] = 1.0
        r[n] = -1.0
        A.append(r)
        lb.append(-np.inf)
        ub.append(-sizes[i])
    for p, (i, j) in enumerate(pairs):
        # o_i + s_i - o_j <= M (1 - y)  ->  o_i - o_j + M y <= M - s_i
        r = np.zeros(nv)
        r[i] = 1.0
        r[j] = -1.0
        r[n + 1 + p] = bigM
        A.append(r)
        lb.append(-np.inf)
        ub.append(bigM - sizes[i])
        # o_j + s_j - o_i <= M y  ->  o_j - o_i - M y <= -s_j
        r = np.zeros(nv)
        r[j] = 1.0
        r[i] = -1.0
        r[n + 1 + p] = -bigM
        A.append(r)
        lb.append(-np.inf)
        ub.append(-sizes[j])

    from scipy.optimize import Bounds
    lo = np.zeros(nv)
    hi = np.full(nv, float(bigM))
    hi[n + 1:] = 1.0
    integrality = np.zeros(nv)
    integrality[:n + 1] = 0
    integrality[n + 1:] = 1

    res = milp(c=c,
               constraints=LinearConstraint(np.array(A), lb, ub),
               bounds=Bounds(lo, hi), integrality=integrality,
               options={"time_limit": time_limit})
    if not res.success:
        logger.info("ilp_pack: solver failed (%s), using skyline",
                    res.message)
        return pack_buffers(buffers)
    addresses = {}
    for i, b in enumerate(buffers):
        addresses[(b.node_name, b.alloc_idx)] = int(round(res.x[i])) \
            // ALIGN * ALIGN
    peak = int(round(res.x[n]))
    sky_addr, sky_peak = pack_buffers(buffers)
    if sky_peak < peak:             # never worse than the heuristic
        return sky_addr, sky_peak
    return addresses, peak
