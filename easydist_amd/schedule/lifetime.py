"""Buffer lifetimes for the static memory plan.

Capability parity with reference ``easydist/torch/schedule/
lifetime_info.py`` (ASAP/ALAP makespans, buffer makespans 151-836),
simplified to the fixed execution order the MI355X runtime actually uses:
the sharded graph runs nodes in list order (one stream for compute;
collectives overlap on side streams but their buffers are allocated on
the compute stream by the *_start wrappers), so a buffer's lifetime is
just [producer index, last consumer index].
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch.fx as fx

from ..memory.allocator_profiler import GraphMemInfo


@dataclass
class Buffer:
    node_name: str
    alloc_idx: int          # which allocation of that node
    size: int
    start: int              # execution index of allocation
    end: int                # last execution index where it may be read
    is_temp: bool


def build_lifetimes(gm: fx.GraphModule, mem_info: GraphMemInfo
                    ) -> List[Buffer]:
    nodes = [n for n in gm.graph.nodes]
    exec_index: Dict[str, int] = {}
    for ni in mem_info.nodes:
        exec_index[ni.name] = ni.index
    n_exec = len(mem_info.nodes)

    # last consumer per node (by name), in execution index space
    last_use: Dict[str, int] = {}
    out_node = nodes[-1]
    for n in nodes:
        for inp in n.all_input_nodes:
            if inp.name in exec_index:
                if n is out_node or n.name not in exec_index:
                    last_use[inp.name] = n_exec    # graph output: keep alive
                else:
                    last_use[inp.name] = max(last_use.get(inp.name, -1),
                                             exec_index[n.name])

    by_name = {ni.name: ni for ni in mem_info.nodes}
    buffers: List[Buffer] = []
    for ni in mem_info.nodes:
        out_allocs = set(i for i in ni.out_alloc_idx if i is not None)
        for i, (size, _stream) in enumerate(ni.allocs):
            if i in out_allocs:
                end = last_use.get(ni.name, ni.index)
                buffers.append(Buffer(ni.name, i, size, ni.index, end,
                                      is_temp=False))
            else:
                # temp workspace: dies when the op finishes
                buffers.append(Buffer(ni.name, i, size, ni.index, ni.index,
                                      is_temp=True))
    return buffers
