"""xGMI/CDNA4 communication & compute cost model for the strategy solver.

Unlike the reference's alpha-free generic bandwidth terms
(reference easydist/autoflow/solver.py:49-113), this model is fit to the
MI355X node fabric: xGMI is point-to-point — each GPU has 7 links of
~153 GB/s to its peers, so RCCL ring collectives are bound by ONE link's
bandwidth per direction, not by an NVSwitch aggregate. All-to-all, done as
simultaneous pairwise exchanges, spreads over all 7 links at once.
"""
from __future__ import annotations

from .. import config as mdconfig
from ..metashard.metair import SPMD


def ring_time(bytes_on_wire: float, n: int) -> float:
    """Time for a ring pass moving `bytes_on_wire` per rank over one link."""
    if n <= 1:
        return 0.0
    return bytes_on_wire / mdconfig.XGMI_LINK_BW + mdconfig.COLLECTIVE_LATENCY


def all_gather_cost(nbytes: float, n: int) -> float:
    # ring all-gather: each rank receives (n-1)/n of the result over 1 link
    return ring_time(nbytes * (n - 1) / n, n)


def all_reduce_cost(nbytes: float, n: int) -> float:
    # reduce-scatter + all-gather
    return ring_time(2 * nbytes * (n - 1) / n, n)


def reduce_scatter_cost(nbytes: float, n: int) -> float:
    return ring_time(nbytes * (n - 1) / n, n)


def all_to_all_cost(nbytes: float, n: int) -> float:
    """Pairwise exchange over xGMI: each rank moves (n-1)/n of its local
    shard, spread over min(n-1, 7) concurrent links."""
    if n <= 1:
        return 0.0
    links = min(n - 1, mdconfig.XGMI_NUM_LINKS)
    per_link = nbytes * (n - 1) / (n * n) / links * (n - 1)
    return (per_link / mdconfig.XGMI_LINK_BW
            + mdconfig.COLLECTIVE_LATENCY) * mdconfig.all_to_all_punish_factor


def reshard_cost(src: SPMD, dst: SPMD, nbytes: float, n: int) -> float:
    """Cost (seconds) of transforming placement src -> dst on a mesh dim of
    size n for a tensor of global size nbytes."""
    if n <= 1:
        return 0.0
    if repr(src) == repr(dst):
        return 0.0
    if src.is_replicate():
        if dst.is_shard():
            return 0.0            # local slice
        if dst.is_partial():
            return 0.0            # divide locally (rare; rank0-keep)
    if src.is_shard():
        if dst.is_replicate():
            return all_gather_cost(nbytes, n)
        if dst.is_shard():
            return all_to_all_cost(nbytes, n)
        if dst.is_partial():
            # S -> P never useful; price as gather
            return all_gather_cost(nbytes, n)
    if src.is_partial():
        if dst.is_replicate():
            return all_reduce_cost(nbytes, n)
        if dst.is_shard():
            return reduce_scatter_cost(nbytes, n)
        if dst.is_partial():
            return all_reduce_cost(nbytes, n)
    return all_reduce_cost(nbytes, n)


def mfma_time(flops: float, dtype_bytes: int = 2) -> float:
    peak = (mdconfig.MFMA_BF16_FLOPS if dtype_bytes <= 2
            else mdconfig.MFMA_FP32_FLOPS)
    return flops / peak


def hbm_time(nbytes: float) -> float:
    return nbytes / mdconfig.HBM_BW
