"""Multi-rank dispatch + stitch covered on CPU: world_size 2 over gloo.

The same shard/stitch code runs under RCCL on the 8xMI355X node (bench.py
--gpus N); here two spawned processes exercise the task partition and the
disjoint-sub-volume gather with CPU tensors."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, result_path):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import (init_distributed, shard_tasks,
                                        stitch_to_rank0)
    r, w = init_distributed(backend='gloo')
    assert (r, w) == (rank, world)

    bboxes = BoundingBoxes.from_manual_setup(
        (4, 8, 8), roi_size=(4, 8, 8 * 4))  # 4 tasks along x
    assert len(bboxes) == 4
    mine = shard_tasks(list(range(len(bboxes))), rank, world)
    assert mine == [i for i in range(4) if i % world == rank]

    local = {}
    for i in mine:
        t = torch.full((3, 4, 8, 8), float(i + 1), dtype=torch.float32)
        local[i] = t
    vol = stitch_to_rank0(bboxes, local, 3, rank, world, 'cpu')
    if rank == 0:
        arr = vol.numpy()
        np.save(result_path, arr)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_shard_and_stitch_world2(tmp_path):
    result = str(tmp_path / 'vol.npy')
    port = 29531
    mp.spawn(_worker, args=(2, port, result), nprocs=2, join=True)
    vol = np.load(result)
    assert vol.shape == (3, 4, 8, 32)
    for i in range(4):
        np.testing.assert_array_equal(vol[:, :, :, i * 8:(i + 1) * 8],
                                      np.full((3, 4, 8, 8), i + 1,
                                              dtype=np.float32))


def test_shard_tasks_partition():
    from chunkflow_amd.dispatch import shard_tasks
    tasks = list(range(10))
    parts = [shard_tasks(tasks, r, 4) for r in range(4)]
    assert sorted(sum(parts, [])) == tasks
    assert parts[0] == [0, 4, 8]
