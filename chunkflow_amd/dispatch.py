"""In-node multi-GPU task dispatch + output-volume stitch.

The reference distributes independent chunk tasks through AWS SQS / slurm
indices with zero inter-task communication. The MI355X-native
equivalent (BASELINE config 3): one process per GPU over torch.distributed
(backend "nccl" = RCCL over xGMI), tasks sharded by index modulo rank, and a
single collective — the final gather of each rank's disjoint output
sub-volume to rank 0, done with point-to-point isend/irecv pairs (disjoint
targets; concurrent p2p over the 7x ~153 GB/s xGMI links beats a ring
collective for a gather of disjoint blocks — SURVEY.md §5).

CPU-coverage: the same code runs under the "gloo" backend with world_size 2
in tests/test_dispatch_gloo.py.
"""
import os
from typing import List, Optional

import torch
import torch.distributed as dist

from .cartesian import BoundingBox, Cartesian


def init_distributed(backend: str = None):
    """Read torchrun's env (RANK/WORLD_SIZE/MASTER_*); single-process if
    absent. Returns (rank, world_size)."""
    world = int(os.environ.get('WORLD_SIZE', '1'))
    if world == 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = os.environ.get(
                'CFX_DIST_BACKEND',
                'nccl' if torch.cuda.is_available() else 'gloo')
        rank = int(os.environ['RANK'])
        if backend == 'nccl':
            torch.cuda.set_device(int(os.environ.get('LOCAL_RANK', rank)))
        dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


def shard_tasks(tasks: list, rank: int, world_size: int) -> list:
    """Task index modulo rank — the reference's slurm-array index selection
    (flow.py:554-582) without the queue."""
    return [t for i, t in enumerate(tasks) if i % world_size == rank]


def stitch_to_rank0(bboxes: List[BoundingBox], local_outputs: dict,
                    channels: int, rank: int, world_size: int,
                    device, roi_start: Cartesian = None,
                    dtype=torch.float32) -> Optional[torch.Tensor]:
    """Gather disjoint per-task output sub-volumes into one volume on rank 0.

    bboxes: ALL task boxes in global index order (every rank knows them);
    local_outputs: {task_index: tensor (C, dz, dy, dx)} owned by this rank
    (task i belongs to rank i % world_size). Returns the stitched
    (C, *roi_shape) tensor on rank 0, None elsewhere.

    P2P matching relies on per-source ordering, not tags (RCCL ignores
    tags): both sides walk the same global task order.

    Backend note: gloo moves CPU tensors only, so under gloo any device
    sub-volumes bounce through host memory and the stitched volume
    returns on the original device — functionally identical, used for
    dress-rehearsing the torchrun path on boxes where RCCL cannot form
    the communicator (e.g. two ranks on one GPU).
    """
    gloo_cuda = (dist.is_initialized() and world_size > 1
                 and dist.get_backend() == 'gloo'
                 and any(t.is_cuda for t in local_outputs.values()))
    if gloo_cuda:
        orig_device = device
        local_outputs = {i: t.cpu() for i, t in local_outputs.items()}
        device = 'cpu'
    if roi_start is None:
        roi_start = Cartesian(*(min(b.start[d] for b in bboxes)
                                for d in range(3)))
    roi_stop = Cartesian(*(max(b.stop[d] for b in bboxes)
                           for d in range(3)))
    roi_shape = tuple(roi_stop - roi_start)

    if world_size == 1:
        volume = torch.zeros((channels,) + roi_shape, dtype=dtype,
                             device=device)
        for i, bbox in enumerate(bboxes):
            s = bbox.start - roi_start
            d = bbox.shape
            volume[:, s.z:s.z + d.z, s.y:s.y + d.y, s.x:s.x + d.x] = \
                local_outputs[i]
        return volume

    if rank == 0:
        volume = torch.zeros((channels,) + roi_shape, dtype=dtype,
                             device=device)
        reqs = []
        staged = {}
        for i, bbox in enumerate(bboxes):
            owner = i % world_size
            if owner == 0:
                s = bbox.start - roi_start
                d = bbox.shape
                volume[:, s.z:s.z + d.z, s.y:s.y + d.y,
                       s.x:s.x + d.x] = local_outputs[i]
            else:
                buf = torch.empty((channels,) + tuple(bbox.shape),
                                  dtype=dtype, device=device)
                staged[i] = buf
                reqs.append(dist.irecv(buf, src=owner))
        for r in reqs:
            r.wait()
        for i, buf in staged.items():
            s = bboxes[i].start - roi_start
            d = bboxes[i].shape
            volume[:, s.z:s.z + d.z, s.y:s.y + d.y, s.x:s.x + d.x] = buf
        return volume.to(orig_device) if gloo_cuda else volume
    else:
        reqs = []
        for i, bbox in enumerate(bboxes):
            if i % world_size == rank:
                t = local_outputs[i]
                if not t.is_contiguous():
                    t = t.contiguous()
                reqs.append(dist.isend(t, dst=0))
        for r in reqs:
            r.wait()
        return None


def all_reduce_hook(group=None):
    """pre_normalize_hook for intra-chunk sharding: SUM the per-rank
    partial blends (one RCCL all-reduce per chunk — the only data-path
    collective the sharded path needs)."""
    def hook(output_tensor):
        dist.all_reduce(output_tensor, op=dist.ReduceOp.SUM, group=group)
    return hook
