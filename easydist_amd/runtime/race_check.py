"""Runtime memory-ownership checker (optional race detector).

Capability parity with reference ``easydist/torch/compile_auto.py``
269-351 (mem_owner_tracer + op_mem_checker under ENABLE_RUNTIME_TRACE):
run the sharded graph under an interpreter that records, per storage
interval, which node last wrote it; before each op executes, verify that
every tensor input's memory was last written by the producer the graph
says — a violation means the memory plan (or an in-place op) clobbered a
live buffer.
"""
from __future__ import annotations

import bisect
import logging
from typing import Dict, List, Optional, Tuple

import torch
import torch.fx as fx

logger = logging.getLogger(__name__)


class _IntervalOwners:
    """Sorted disjoint [start, end) -> writer name."""

    def __init__(self):
        self.starts: List[int] = []
        self.ivals: List[Tuple[int, int, str]] = []

    def write(self, start: int, end: int, writer: str):
        # remove/trim any overlapping intervals, then insert
        i = bisect.bisect_right(self.starts, start) - 1
        keep: List[Tuple[int, int, str]] = []
        j = max(i, 0)
        while j < len(self.ivals):
            s, e, w = self.ivals[j]
            if s >= end:
                break
            if e <= start:
                j += 1
                continue
            if s < start:
                keep.append((s, start, w))
            if e > end:
                keep.append((end, e, w))
            self.ivals.pop(j)
            self.starts.pop(j)
        for s, e, w in keep + [(start, end, writer)]:
            p = bisect.bisect_left(self.starts, s)
            self.starts.insert(p, s)
            self.ivals.insert(p, (s, e, w))

    def owner(self, start: int, end: int) -> Optional[str]:
        """Single owner covering [start, end), else None."""
        i = bisect.bisect_right(self.starts, start) - 1
        if i < 0 or i >= len(self.ivals):
            return None
        s, e, w = self.ivals[i]
        if s <= start and end <= e:
            return w
        return None


def _tensors(v):
    out = []

    def rec(x):
        if isinstance(x, torch.Tensor):
            out.append(x)
        elif isinstance(x, (list, tuple)):
            for y in x:
                rec(y)
    rec(v)
    return out


class MemOwnershipChecker(fx.Interpreter):
    """Interpret the graph once, checking producer/consumer memory
    ownership. Returns the list of violations."""

    def __init__(self, gm: fx.GraphModule):
        super().__init__(gm)
        self.owners = _IntervalOwners()
        self.producer_of: Dict[int, str] = {}   # id(tensor) -> node name
        self.violations: List[str] = []

    def run_node(self, n: fx.Node):
        if n.op == "call_function":
            for a in n.all_input_nodes:
                v = self.env.get(a)
                for t in _tensors(v):
                    if not t.is_cuda and t.device.type != "cpu":
                        continue
                    start = t.data_ptr()
                    end = start + t.numel() * t.element_size()
                    owner = self.owners.owner(start, end)
                    expect = self.producer_of.get(id(t))
                    if owner is not None and expect is not None \
                            and owner != expect:
                        self.violations.append(
                            f"{n.name}: input from {a.name} was last "
                            f"written by {owner}, expected {expect}")
        result = super().run_node(n)
        if n.op in ("call_function", "placeholder"):
            for t in _tensors(result):
                start = t.data_ptr()
                end = start + t.numel() * t.element_size()
                self.owners.write(start, end, n.name)
                self.producer_of[id(t)] = n.name
        return result

    def check(self, args) -> List[str]:
        self.run(*args)
        for v in self.violations:
            logger.warning("memory ownership violation: %s", v)
        return self.violations
