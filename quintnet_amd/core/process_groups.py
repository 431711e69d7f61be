"""Process-group manager: user-facing init of the [dp, tp, pp] mesh.

Parity with reference core/process_groups.py:42-181 (`ProcessGroupManager`,
`init_process_groups`).  MI355X-native notes: axis subgroups are RCCL
communicators created once at startup; the manager also hands out the
axis rank/size/coords used by the TP/PP/DP engines.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist

from .mesh import MeshGenerator

__all__ = ["ProcessGroupManager", "init_process_groups"]


class ProcessGroupManager:
    def __init__(
        self,
        device_type: str = "cuda",
        mesh_dim: Sequence[int] = (2, 2, 2),
        mesh_name: Sequence[str] = ("dp", "tp", "pp"),
        timeout_s: float = 600.0,
    ) -> None:
        mesh_dim = list(mesh_dim)
        mesh_name = list(mesh_name)
        if len(mesh_dim) != len(mesh_name):
            raise ValueError("mesh_dim and mesh_name must have equal length")
        n = math.prod(mesh_dim)
        mesh = torch.arange(n).view(mesh_dim)
        self.mesh_dim = mesh_dim
        self.mesh_name = mesh_name
        self.device_type = device_type
        self.mesh_generator = MeshGenerator(device_type, mesh, mesh_name, timeout_s)
        self.mesh = self.mesh_generator.mesh

    # -- lookup ---------------------------------------------------------
    @property
    def rank(self) -> int:
        return self.mesh_generator.rank

    @property
    def world_size(self) -> int:
        return self.mesh_generator.world_size

    @property
    def device(self) -> torch.device:
        return self.mesh_generator.device

    def get_group(self, name: str) -> dist.ProcessGroup:
        return self.mesh_generator.get_group(name)

    def get_all_groups(self) -> Dict[str, dist.ProcessGroup]:
        return dict(self.mesh_generator.groups)

    def get_group_ranks(self, name: str) -> List[int]:
        return self.mesh_generator.get_group_ranks(name)

    def get_coordinates_tensor_search(self, rank: Optional[int] = None) -> List[int]:
        return self.mesh_generator.get_coordinates(rank)

    def get_tied_embedding_group(self):
        """first+last-pp-stage subgroup for tied-weight grad sync."""
        return self.mesh_generator.tied_embedding_group

    get_coordinates = get_coordinates_tensor_search

    # -- axis helpers ---------------------------------------------------
    def axis_size(self, name: str) -> int:
        if name not in self.mesh_name:
            return 1
        return self.mesh_dim[self.mesh_name.index(name)]

    def axis_rank(self, name: str) -> int:
        if name not in self.mesh_name:
            return 0
        return self.get_coordinates()[self.mesh_name.index(name)]

    # common shorthands
    @property
    def dp_size(self) -> int:
        return self.axis_size("dp")

    @property
    def tp_size(self) -> int:
        return self.axis_size("tp")

    @property
    def pp_size(self) -> int:
        return self.axis_size("pp")

    @property
    def dp_rank(self) -> int:
        return self.axis_rank("dp")

    @property
    def tp_rank(self) -> int:
        return self.axis_rank("tp")

    @property
    def pp_rank(self) -> int:
        return self.axis_rank("pp")

    def print_mesh_info(self) -> None:
        if self.rank == 0:
            print(f"[quintnet_amd] mesh {self.mesh_dim} axes {self.mesh_name}")
            print(self.mesh)
        for name in self.mesh_name:
            if self.rank == 0:
                print(f"  axis '{name}': size {self.axis_size(name)}")


def init_process_groups(
    device_type: str = "cuda",
    mesh_dim: Sequence[int] = (2, 2, 2),
    mesh_name: Sequence[str] = ("dp", "tp", "pp"),
    timeout_s: float = 600.0,
) -> ProcessGroupManager:
    """Initialize torch.distributed (RCCL/gloo) + the axis subgroups.

    Reference parity: core/process_groups.py:163-181.
    """
    return ProcessGroupManager(device_type, mesh_dim, mesh_name, timeout_s)
