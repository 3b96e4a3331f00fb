"""Device mesh & parallelism degrees for one-process-per-GPU over RCCL.

Capability parity with the reference device mesh (reference:
src/modalities/running_env/fsdp/device_mesh.py:84-215) but implemented as an
explicit process-group lattice: ranks factor as
(pp, dp_replicate, dp_shard, cp, tp) row-major, and each dimension gets its
own RCCL communicator. Intra-node xGMI is fully connected (7x153 GB/s
point-to-point links), so TP/CP groups should stay intra-node and DP-shard
can span the node."""

from dataclasses import dataclass
from enum import Enum
from typing import Optional

import torch.distributed as dist


class ParallelismDegrees(str, Enum):
    PP = "pp"
    DP_REPLICATE = "dp_replicate"
    DP_SHARD = "dp_shard"
    CP = "cp"
    TP = "tp"


@dataclass
class MeshDim:
    name: str
    size: int
    rank: int                  # this process's coordinate in the dim
    group: Optional[object]    # ProcessGroup or None when size == 1

    @property
    def degree(self) -> int:
        return self.size


class DeviceMesh:
    """Lattice of process groups over the world.

    Dimension order (outer->inner): pp, dp_replicate, dp_shard, cp, tp.
    Inner dimensions have stride 1 in global rank — keep tp innermost so TP
    groups are consecutive local GPUs on one xGMI-connected node."""

    DIM_ORDER = [ParallelismDegrees.PP, ParallelismDegrees.DP_REPLICATE,
                 ParallelismDegrees.DP_SHARD, ParallelismDegrees.CP,
                 ParallelismDegrees.TP]

    def __init__(self, world_size: int, rank: int,
                 pp: int = 1, dp_replicate: int = 1, dp_shard: int = -1,
                 cp: int = 1, tp: int = 1, create_groups: bool = True):
        sizes = {ParallelismDegrees.PP: pp, ParallelismDegrees.DP_REPLICATE: dp_replicate,
                 ParallelismDegrees.CP: cp, ParallelismDegrees.TP: tp}
        known = pp * dp_replicate * cp * tp
        if dp_shard in (-1, None):
            if world_size % known != 0:
                raise ValueError(f"world_size {world_size} not divisible by "
                                 f"pp*dp_replicate*cp*tp={known}")
            dp_shard = world_size // known
        sizes[ParallelismDegrees.DP_SHARD] = dp_shard
        total = known * dp_shard
        if total != world_size:
            raise ValueError(f"Mesh {sizes} does not cover world_size {world_size}")
        self.world_size = world_size
        self.rank = rank
        self.sizes = sizes
        self.dims: dict[ParallelismDegrees, MeshDim] = {}
        self._build(create_groups and world_size > 1 and dist.is_initialized())

    def _coords(self, rank: int) -> dict:
        coords = {}
        rem = rank
        for dim in reversed(self.DIM_ORDER):  # innermost first
            size = self.sizes[dim]
            coords[dim] = rem % size
            rem //= size
        return coords

    def _build(self, create_groups: bool):
        my_coords = self._coords(self.rank)
        for dim in self.DIM_ORDER:
            size = self.sizes[dim]
            group = None
            if create_groups and size > 1:
                # Partition all ranks into groups that differ only in `dim`.
                groups: dict[tuple, list[int]] = {}
                for r in range(self.world_size):
                    c = self._coords(r)
                    key = tuple(c[d] for d in self.DIM_ORDER if d != dim)
                    groups.setdefault(key, []).append(r)
                for key, ranks in sorted(groups.items()):
                    g = dist.new_group(ranks=ranks)
                    if self.rank in ranks:
                        group = g
            self.dims[dim] = MeshDim(dim.value, size, my_coords[dim], group)

    # -- query api (reference: device_mesh.py:148-215) --------------------
    def get_degree(self, dim: ParallelismDegrees) -> int:
        return self.sizes[dim]

    def get_rank(self, dim: ParallelismDegrees) -> int:
        return self.dims[dim].rank

    def get_group(self, dim: ParallelismDegrees):
        return self.dims[dim].group

    @property
    def dp_degree(self) -> int:
        return (self.get_degree(ParallelismDegrees.DP_REPLICATE)
                * self.get_degree(ParallelismDegrees.DP_SHARD))

    @property
    def dp_rank(self) -> int:
        return (self.get_rank(ParallelismDegrees.DP_REPLICATE)
                * self.get_degree(ParallelismDegrees.DP_SHARD)
                + self.get_rank(ParallelismDegrees.DP_SHARD))


def get_device_mesh(world_size: Optional[int] = None, rank: Optional[int] = None,
                    pipeline_parallel_degree: int = 1,
                    data_parallel_replicate_degree: int = 1,
                    data_parallel_shard_degree: int = -1,
                    context_parallel_degree: int = 1,
                    tensor_parallel_degree: int = 1) -> DeviceMesh:
    if world_size is None:
        world_size = dist.get_world_size() if dist.is_initialized() else 1
    if rank is None:
        rank = dist.get_rank() if dist.is_initialized() else 0
    return DeviceMesh(world_size, rank,
                      pp=pipeline_parallel_degree,
                      dp_replicate=data_parallel_replicate_degree,
                      dp_shard=data_parallel_shard_degree,
                      cp=context_parallel_degree,
                      tp=tensor_parallel_degree)
