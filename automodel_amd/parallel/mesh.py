"""Device-mesh construction and process-group init for RCCL over xGMI.

Reference behavior: nemo_automodel/components/distributed/mesh_utils.py:286-425
(single named root mesh (pp, dp_replicate, dp_shard, cp, tp) + flattened axes
dp, dp_shard_cp, dp_cp). On MI355X the backend string is "nccl" (RCCL is the
NCCL implementation on ROCm); one process per GPU over xGMI.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass, field

import torch
import torch.distributed as dist
from torch.distributed.device_mesh import DeviceMesh, init_device_mesh


def init_distributed(timeout_minutes: int = 30) -> tuple[int, int, int]:
    """init_process_group from torchrun env; returns (rank, local_rank, world)."""
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1 and not dist.is_initialized():
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(minutes=timeout_minutes)
        )
    elif torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world


@dataclass
class MeshContext:
    """Holds the root mesh and the derived sub-meshes / flattened axes."""

    mesh: DeviceMesh | None
    world_size: int = 1
    dims: dict = field(default_factory=dict)   # name -> size

    @property
    def dp_size(self) -> int:
        return self.dims.get("dp_replicate", 1) * self.dims.get("dp_shard", 1)

    @property
    def dp_cp_size(self) -> int:
        return self.dp_size * self.dims.get("cp", 1)

    @property
    def tp_size(self) -> int:
        return self.dims.get("tp", 1)

    @property
    def pp_size(self) -> int:
        return self.dims.get("pp", 1)

    def __getitem__(self, name: str) -> DeviceMesh:
        assert self.mesh is not None, "single-process run has no mesh"
        return self.mesh[name]

    def get(self, name: str):
        if self.mesh is None:
            return None
        return self.mesh[name]

    @property
    def dp_rank(self) -> int:
        if self.mesh is None:
            return 0
        return self.mesh["dp"].get_local_rank()

    def dp_group(self):
        if self.mesh is None:
            return None
        return self.mesh["dp"].get_group()


def build_mesh(
    dp_replicate: int = 1,
    dp_shard: int = -1,
    tp: int = 1,
    pp: int = 1,
    cp: int = 1,
    device_type: str | None = None,
    axis_timeouts: dict | None = None,
) -> MeshContext:
    """Build the named root mesh (pp, dp_replicate, dp_shard, cp, tp) and
    flatten dp / dp_shard_cp / dp_cp, mirroring the reference's axis names.

    ``axis_timeouts``: per-axis collective timeout overrides in minutes
    (reference mesh_utils.py:173 _nccl_backend_override — e.g. a long pp
    timeout for uneven pipeline stages, short dp for fast failure
    detection).  Applied to the axis process groups after construction."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    if dp_shard == -1:
        denom = dp_replicate * tp * pp * cp
        assert world % denom == 0, f"world {world} not divisible by {denom}"
        dp_shard = world // denom
    total = pp * dp_replicate * dp_shard * cp * tp
    assert total == world, f"mesh {total} != world {world}"

    dims = {"pp": pp, "dp_replicate": dp_replicate, "dp_shard": dp_shard, "cp": cp, "tp": tp}
    if world == 1:
        return MeshContext(mesh=None, world_size=1, dims=dims)

    if device_type is None:
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    names, sizes = zip(*[(k, v) for k, v in dims.items()])
    mesh = init_device_mesh(device_type, sizes, mesh_dim_names=names)
    mesh[("dp_replicate", "dp_shard")]._flatten("dp")
    mesh[("dp_replicate", "dp_shard", "cp")]._flatten("dp_cp")
    mesh[("dp_shard", "cp")]._flatten("dp_shard_cp")
    if axis_timeouts:
        import datetime

        from torch.distributed.distributed_c10d import _set_pg_timeout

        for axis, minutes in axis_timeouts.items():
            if dims.get(axis, 1) > 1:
                _set_pg_timeout(datetime.timedelta(minutes=float(minutes)),
                                mesh[axis].get_group())
    return MeshContext(mesh=mesh, world_size=world, dims=dims)
