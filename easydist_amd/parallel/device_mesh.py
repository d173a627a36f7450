"""NDDeviceMesh: named-dim device mesh over torch DeviceMesh.

Capability parity with reference ``easydist/torch/device_mesh.py``
(NDDeviceMesh wrapper, named dims, 'spmd' binding, submesh coords; lines
31-176). On ROCm the "nccl" backend IS RCCL over xGMI.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

_device_mesh: Optional["NDDeviceMesh"] = None


class NDDeviceMesh:
    def __init__(self, mesh, dim_names: Optional[Sequence[str]] = None):
        """mesh: torch.distributed.device_mesh.DeviceMesh or int tensor/list."""
        from torch.distributed.device_mesh import DeviceMesh
        if isinstance(mesh, DeviceMesh):
            self.mesh = mesh
            self.dim_names = list(mesh.mesh_dim_names or
                                  [f"spmd{i}" for i in range(mesh.ndim)])
        else:
            t = torch.as_tensor(mesh)
            names = list(dim_names) if dim_names else [f"spmd{i}"
                                                       for i in range(t.dim())]
            device_type = "cuda" if torch.cuda.is_available() else "cpu"
            self.mesh = DeviceMesh(device_type, t, mesh_dim_names=tuple(names))
            self.dim_names = names

    # ------------------------------------------------------------ topology ---
    @property
    def ndim(self) -> int:
        return self.mesh.ndim

    @property
    def shape(self) -> List[int]:
        return list(self.mesh.mesh.shape)

    def size(self, dim: Optional[int] = None) -> int:
        if dim is None:
            return self.mesh.mesh.numel()
        return self.mesh.mesh.shape[dim]

    def my_coords(self) -> List[int]:
        rank = dist.get_rank() if dist.is_initialized() else 0
        loc = (self.mesh.mesh == rank).nonzero()
        if loc.numel() == 0:
            return [0] * self.ndim
        return [int(x) for x in loc[0]]

    def get_group(self, dim: int):
        """The process group spanning mesh dim `dim` at this rank's coords."""
        return self.mesh.get_group(dim)

    def spmd_dims(self) -> List[int]:
        """Indices of the 'spmd*' dims ('spmd' binding of the reference)."""
        out = [i for i, n in enumerate(self.dim_names) if n.startswith("spmd")]
        return out if out else list(range(self.ndim))

    def has_dim(self, name: str) -> bool:
        return name in self.dim_names

    def dim_index(self, name: str) -> int:
        return self.dim_names.index(name)

    def __repr__(self):
        return f"NDDeviceMesh(shape={self.shape}, dims={self.dim_names})"


def set_device_mesh(mesh, dim_names: Optional[Sequence[str]] = None):
    global _device_mesh
    if isinstance(mesh, NDDeviceMesh):
        _device_mesh = mesh
    else:
        _device_mesh = NDDeviceMesh(mesh, dim_names)
    return _device_mesh


def get_device_mesh() -> Optional[NDDeviceMesh]:
    return _device_mesh
