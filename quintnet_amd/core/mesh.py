"""N-dimensional device mesh over torch.distributed (RCCL on ROCm).

Builds the rank grid for the ``[dp, tp, pp]`` axes and one communicator
subgroup per mesh axis per rank.  Functional parity with the reference
mesh generator (reference: core/mesh.py:132-294) but designed for a
single MI355X node: the 8 GPUs are fully connected over xGMI (7
point-to-point links per GPU), so axis subgroups are just RCCL
communicators — no topology-aware reordering is required for
correctness, and the 2-wide groups used by the [2,2,2] mesh each map to
a single dedicated xGMI link.

Unlike the reference, subgroup creation is world-uniform: every rank
enumerates and calls ``dist.new_group`` for *every* row of every axis
(reference only loops its own rows, which happens to work but is
fragile — see SURVEY.md §3.1).
"""

from __future__ import annotations

import datetime
import os
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist

__all__ = ["MeshGenerator"]


def _default_backend(device_type: str) -> str:
    # "nccl" IS RCCL on ROCm builds of PyTorch.
    if device_type == "cuda" and torch.cuda.is_available():
        return "nccl"
    return "gloo"


class MeshGenerator:
    """Create per-axis process groups from an N-D mesh tensor.

    Args:
        device_type: "cuda" (MI355X via ROCm/HIP) or "cpu" (tests, gloo).
        mesh: integer tensor of shape ``mesh_dim`` holding global ranks.
        mesh_dim_names: one name per mesh axis, e.g. ``("dp","tp","pp")``.
    """

    def __init__(
        self,
        device_type: str,
        mesh: torch.Tensor,
        mesh_dim_names: Sequence[str],
        timeout_s: float = 600.0,
    ) -> None:
        if mesh.ndim != len(mesh_dim_names):
            raise ValueError(
                f"mesh has {mesh.ndim} dims but {len(mesh_dim_names)} names given"
            )
        self.device_type = device_type
        self.mesh = mesh.to(dtype=torch.long, device="cpu")
        self.mesh_dim_names = tuple(mesh_dim_names)
        self.timeout = datetime.timedelta(seconds=timeout_s)

        self._setup_group_and_device()
        self.groups: Dict[str, dist.ProcessGroup] = {}
        self.group_ranks: Dict[str, List[int]] = {}
        self._init_process_groups()

    # ------------------------------------------------------------------
    def _setup_group_and_device(self) -> None:
        if not dist.is_initialized():
            backend = _default_backend(self.device_type)
            # single-process fallback (tests, bench --gpus 1 without torchrun)
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29571")
            os.environ.setdefault("RANK", "0")
            os.environ.setdefault("WORLD_SIZE", "1")
            dist.init_process_group(backend=backend, timeout=self.timeout)
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        if self.mesh.numel() != self.world_size:
            raise ValueError(
                f"mesh has {self.mesh.numel()} entries but world_size is "
                f"{self.world_size}"
            )
        if self.device_type == "cuda" and torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", self.rank % max(torch.cuda.device_count(), 1)))
            torch.cuda.set_device(local_rank)
            self.device = torch.device("cuda", local_rank)
        else:
            self.device = torch.device("cpu")

    # ------------------------------------------------------------------
    def _init_process_groups(self) -> None:
        """One subgroup per axis per rank; world-uniform enumeration.

        For axis ``d``: move that axis last, flatten the rest; every row
        is one subgroup.  All ranks create all groups (required for
        correctness of ``dist.new_group``); each rank stores the group
        it belongs to.
        """
        for dim, name in enumerate(self.mesh_dim_names):
            size = self.mesh.shape[dim]
            rows = self.mesh.swapdims(-1, dim).reshape(-1, size)
            my_group: Optional[dist.ProcessGroup] = None
            my_ranks: Optional[List[int]] = None
            for row in rows:
                ranks = row.tolist()
                grp = dist.new_group(ranks=ranks, timeout=self.timeout)
                if self.rank in ranks:
                    my_group = grp
                    my_ranks = ranks
            assert my_group is not None and my_ranks is not None
            self.groups[name] = my_group
            self.group_ranks[name] = my_ranks

        # Dedicated duplicate pp communicators for the interleaved-1F1B
        # schedule: its ring P2P wraps around (rank p-1 -> 0 between model
        # chunks) and at pp=2 the forward and backward directions share a
        # rank pair, so each direction gets its OWN communicator to keep
        # per-pair FIFO message matching independent per direction.
        if "pp" in self.mesh_dim_names:
            dim = self.mesh_dim_names.index("pp")
            size = self.mesh.shape[dim]
            rows = self.mesh.swapdims(-1, dim).reshape(-1, size)
            for tag in ("pp_fwd", "pp_bwd"):
                my_group = None
                my_ranks = None
                for row in rows:
                    ranks = row.tolist()
                    grp = dist.new_group(ranks=ranks, timeout=self.timeout)
                    if self.rank in ranks:
                        my_group = grp
                        my_ranks = ranks
                self.groups[tag] = my_group
                self.group_ranks[tag] = my_ranks

        # Tied-embedding subgroup: first+last rank of each pp row (used
        # for GPT-2 wte/lm_head grad sync; restricting the collective to
        # the two stages that own the weight avoids the pp>2 deadlock of
        # the reference's whole-group AVG — SURVEY.md §8.3).
        self.tied_embedding_group = None
        if "pp" in self.mesh_dim_names:
            dim = self.mesh_dim_names.index("pp")
            size = self.mesh.shape[dim]
            if size == 2:
                self.tied_embedding_group = self.groups["pp"]
            elif size > 2:
                rows = self.mesh.swapdims(-1, dim).reshape(-1, size)
                for row in rows:
                    pair = [int(row[0]), int(row[-1])]
                    grp = dist.new_group(ranks=pair, timeout=self.timeout)
                    if self.rank in pair:
                        self.tied_embedding_group = grp

    # ------------------------------------------------------------------
    def get_group(self, name: str) -> dist.ProcessGroup:
        return self.groups[name]

    def get_group_ranks(self, name: str) -> List[int]:
        return self.group_ranks[name]

    def get_coordinates(self, rank: Optional[int] = None) -> List[int]:
        """Mesh coordinates of ``rank`` (defaults to this rank)."""
        if rank is None:
            rank = self.rank
        idx = (self.mesh == rank).nonzero(as_tuple=False)
        if idx.numel() == 0:
            raise ValueError(f"rank {rank} not in mesh")
        return idx[0].tolist()

    # Name kept for API parity with the reference (core/mesh.py:268-294).
    get_coordinates_tensor_search = get_coordinates
