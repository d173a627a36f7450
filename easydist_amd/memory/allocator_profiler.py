"""Allocation profiler pass: classify every buffer of the sharded graph.

Capability parity with reference ``easydist/torch/passes/
allocator_profiler.py`` (AllocatorProfiler 118-306) and
``mem_allocation_info.py``: run the graph once under PROFILE mode setting
the op name per node, then correlate the recorded (op, ptr, size, stream)
tuples with each node's output data_ptrs to classify OUT vars vs TEMP
buffers vs IN-PLACE references, producing a GraphMemInfo the scheduler
packs.
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch
import torch.fx as fx

from . import meta_allocator as ma

logger = logging.getLogger(__name__)


@dataclass
class NodeMemInfo:
    name: str
    index: int                        # execution position
    # (size, stream) per allocation made while this op ran
    allocs: List[Tuple[int, int]] = field(default_factory=list)
    # which of those allocations back this node's outputs: alloc_idx list
    out_alloc_idx: List[Optional[int]] = field(default_factory=list)
    # outputs that alias an input (in-place / view): out position -> True
    inplace_out: List[bool] = field(default_factory=list)


@dataclass
class GraphMemInfo:
    nodes: List[NodeMemInfo]
    # total allocation count in execution order (the plan's malloc order)
    alloc_order: List[Tuple[str, int, int]]   # (node name, alloc idx, size)


class AllocatorProfiler(fx.Interpreter):
    """Run the graph once under PROFILE mode; build GraphMemInfo."""

    def __init__(self, gm: fx.GraphModule):
        super().__init__(gm)
        self.mem_info: List[NodeMemInfo] = []
        self._idx = 0

    def run_node(self, n: fx.Node):
        c = ma.ctl()
        if n.op in ("placeholder", "output", "get_attr"):
            return super().run_node(n)
        input_ptrs = set()
        for a in n.all_input_nodes:
            v = self.env.get(a)
            for t in _tensors(v):
                input_ptrs.add(t.data_ptr())
        c.set_cur_op_name(n.name)
        before = len(c.get_records())
        result = super().run_node(n)
        torch.cuda.synchronize()
        records = c.get_records()[before:]

        info = NodeMemInfo(name=n.name, index=self._idx)
        self._idx += 1
        ptr_to_alloc = {}
        for i, (op, ptr, size, stream) in enumerate(records):
            info.allocs.append((size, stream))
            ptr_to_alloc[ptr] = i
        for t in _tensors(result):
            base = t.data_ptr()
            matched = None
            for ptr, i in ptr_to_alloc.items():
                if ptr <= base < ptr + info.allocs[i][0]:
                    matched = i
                    break
            info.out_alloc_idx.append(matched)
            info.inplace_out.append(matched is None
                                    and base in input_ptrs)
        self.mem_info.append(info)
        return result

    def profile(self, args) -> GraphMemInfo:
        c = ma.ctl()
        c.clear_records()
        c.set_mode(ma.PROFILE)
        try:
            self.run(*args)
        finally:
            c.set_mode(ma.PASSTHROUGH)
        order = []
        for ni in self.mem_info:
            for i, (size, _stream) in enumerate(ni.allocs):
                order.append((ni.name, i, size))
        return GraphMemInfo(nodes=self.mem_info, alloc_order=order)


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
