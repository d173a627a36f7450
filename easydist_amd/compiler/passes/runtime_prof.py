"""Per-op runtime microbenchmark + persistent PerfDB.

Capability parity with reference ``easydist/torch/passes/runtime_prof.py``
(HIP-event timing, 2 warmup + 5 trials, lines 36-131) and
``graph_profile_db.py`` (pickled per-op latency cache). Timings feed the
RCPSP comm scheduler and the solver's overlap discount.
"""
from __future__ import annotations

import hashlib
import logging
import os
import pickle
import time
from typing import Dict, Optional

import torch
import torch.fx as fx

logger = logging.getLogger(__name__)

WARMUP, TRIALS = 2, 5


class PerfDB:
    """(op signature) -> milliseconds, persisted under ~/.easydist_amd."""

    def __init__(self, path: Optional[str] = None):
        self.path = path or os.path.join(
            os.path.expanduser("~"), ".easydist_amd", "perf.db")
        self._db: Dict[str, float] = {}
        if os.path.exists(self.path):
            try:
                with open(self.path, "rb") as f:
                    self._db = pickle.load(f)
            except Exception:
                self._db = {}

    def get(self, key: str) -> Optional[float]:
        return self._db.get(key)

    def put(self, key: str, ms: float):
        self._db[key] = ms

    def save(self):
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        with open(self.path, "wb") as f:
            pickle.dump(self._db, f)


def node_signature(n: fx.Node) -> str:
    shapes = []
    for a in n.all_input_nodes:
        v = a.meta.get("val")
        if isinstance(v, torch.Tensor):
            shapes.append((tuple(v.shape), str(v.dtype)))
    raw = f"{n.target}|{shapes}"
    return hashlib.sha1(raw.encode()).hexdigest()[:16]


class RuntimeProfiler(fx.Interpreter):
    """Run the graph once, timing each call_function node."""

    def __init__(self, gm: fx.GraphModule, db: Optional[PerfDB] = None):
        super().__init__(gm)
        self.db = db or PerfDB()
        self.durations: Dict[str, float] = {}

    def run_node(self, n: fx.Node):
        if n.op != "call_function":
            return super().run_node(n)
        sig = node_signature(n)
        cached = self.db.get(sig)
        if cached is not None:
            self.durations[n.name] = cached
            return super().run_node(n)
        use_events = torch.cuda.is_available()
        args, kwargs = self.fetch_args_kwargs_from_env(n)
        for _ in range(WARMUP):
            n.target(*args, **kwargs)
        if use_events:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            for _ in range(TRIALS):
                result = n.target(*args, **kwargs)
            end.record()
            end.synchronize()
            ms = start.elapsed_time(end) / TRIALS
        else:
            t0 = time.perf_counter()
            for _ in range(TRIALS):
                result = n.target(*args, **kwargs)
            ms = (time.perf_counter() - t0) * 1000.0 / TRIALS
        self.durations[n.name] = ms
        self.db.put(sig, ms)
        self.env[n] = result
        return result

    def profile(self, args) -> Dict[str, float]:
        self.run(*args)
        self.db.save()
        return dict(self.durations)
