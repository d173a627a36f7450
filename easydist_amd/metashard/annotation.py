"""Shard annotations: which input dims may be sharded, grouped by shard-dim id.

Capability parity with the reference's ``easydist/metashard/annotation.py``
(ShardDim/ShardAnnotation, reference lines 22-135); re-designed torch-only.

A ``ShardAnnotation`` assigns every dim of every tensor input a ``ShardDim``.
Dims carrying the same positive ``shard_dim_id`` must be sharded *together*
(e.g. the contraction dims of a matmul); id 0 means "not shardable here".
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List


@dataclass(frozen=True)
class ShardDim:
    shard_dim_id: int
    chunk: int = 1   # block-cyclic chunking factor (1 = plain contiguous chunks)
    halo: int = 0    # overlap elements on each shard boundary (stencil ops)

    @staticmethod
    def get_noshard_dim() -> "ShardDim":
        return ShardDim(0)

    @staticmethod
    def get_shard_dim(idx: int, chunk: int = 1, halo: int = 0) -> "ShardDim":
        return ShardDim(idx, chunk=chunk, halo=halo)

    def __repr__(self):
        if self.shard_dim_id == 0:
            return "NoShardDim"
        s = f"ShardDim({self.shard_dim_id}"
        if self.chunk != 1:
            s += f", chunk={self.chunk}"
        if self.halo != 0:
            s += f", halo={self.halo}"
        return s + ")"


NoShardDim = ShardDim.get_noshard_dim()


@dataclass
class ShardAnnotation:
    """Per-input, per-dim ShardDim assignment."""
    annotation: List[List[ShardDim]] = field(default_factory=list)

    @staticmethod
    def init_from_input_shapes(shapes) -> "ShardAnnotation":
        return ShardAnnotation([[NoShardDim for _ in range(len(s))] for s in shapes])

    def inject_haloinfo(self, halo: int, shard_dim_id: int) -> "ShardAnnotation":
        new = [[ShardDim(sd.shard_dim_id, sd.chunk, halo)
                if sd.shard_dim_id == shard_dim_id else sd for sd in dims]
               for dims in self.annotation]
        return ShardAnnotation(new)

    def get_max_shard_dim_id(self) -> int:
        mx = 0
        for dims in self.annotation:
            for sd in dims:
                mx = max(mx, sd.shard_dim_id)
        return mx

    def positions_of(self, shard_dim_id: int):
        """All (input_idx, dim) pairs carrying this shard_dim_id."""
        out = []
        for i, dims in enumerate(self.annotation):
            for d, sd in enumerate(dims):
                if sd.shard_dim_id == shard_dim_id:
                    out.append((i, d))
        return out

    def clear_shard_dim(self, above_id: int) -> "ShardAnnotation":
        """Drop every assignment with id > above_id."""
        new = [[sd if sd.shard_dim_id <= above_id else NoShardDim for sd in dims]
               for dims in self.annotation]
        return ShardAnnotation(new)

    def copy(self) -> "ShardAnnotation":
        return ShardAnnotation([list(dims) for dims in self.annotation])

    def __getitem__(self, i):
        return self.annotation[i]

    def __setitem__(self, i, v):
        self.annotation[i] = v

    def __len__(self):
        return len(self.annotation)

    def __repr__(self):
        return f"ShardAnnotation({self.annotation})"
