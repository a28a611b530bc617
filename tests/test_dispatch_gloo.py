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


def _intra_worker(rank, world, port, result_path):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import numpy as _np
    import torch.distributed as dist
    from chunkflow_amd.chunk import Chunk
    from chunkflow_amd.dispatch import init_distributed, all_reduce_hook
    from chunkflow_amd.inferencer import Inferencer
    init_distributed(backend='gloo')
    arr = _np.load(result_path + '.in.npy')
    inf = Inferencer(None, None, (10, 32, 32),
                     output_patch_overlap=(2, 8, 8), framework='identity',
                     num_output_channels=2, batch_size=3,
                     mask_output_chunk=True,
                     patch_shard=(rank, world),
                     pre_normalize_hook=all_reduce_hook())
    out = inf(Chunk(arr))
    if rank == 0:
        _np.save(result_path, out.numpy().array)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_intra_chunk_shard_world2(tmp_path):
    """Intra-chunk patch-grid sharding: 2 ranks
    share one chunk's patches; the all-reduced result equals the
    single-rank oracle at 1e-5."""
    from oracle import oracle_inference
    rng = np.random.RandomState(17)
    arr = rng.randint(0, 256, size=(20, 68, 72), dtype=np.uint8)
    result = str(tmp_path / 'intra.npy')
    np.save(result + '.in.npy', arr)
    mp.spawn(_intra_worker, args=(2, 29533, result), nprocs=2, join=True)
    got = np.load(result)
    ref = oracle_inference(arr, (10, 32, 32), (2, 8, 8),
                           num_output_channels=2, batch_size=3)
    np.testing.assert_allclose(got, ref, rtol=1e-5, atol=1e-6)


@pytest.mark.timeout(240)
def test_shard_and_stitch_world4(tmp_path):
    """World 4 — the same fan-in rank 0 sees on the 8-GPU node (every
    non-zero rank sends concurrently)."""
    result = str(tmp_path / 'vol4.npy')
    mp.spawn(_worker, args=(4, 29537, result), nprocs=4, join=True)
    vol = np.load(result)
    assert vol.shape == (3, 4, 8, 32)
    for i in range(4):
        np.testing.assert_array_equal(vol[:, :, :, i * 8:(i + 1) * 8],
                                      np.full((3, 4, 8, 8), i + 1,
                                              dtype=np.float32))


@pytest.mark.timeout(240)
def test_intra_chunk_shard_world3(tmp_path):
    """Intra-chunk sharding with a world size that does NOT divide the
    patch count evenly (ragged shard)."""
    from oracle import oracle_inference
    rng = np.random.RandomState(23)
    arr = rng.randint(0, 256, size=(20, 68, 72), dtype=np.uint8)
    result = str(tmp_path / 'intra3.npy')
    np.save(result + '.in.npy', arr)
    mp.spawn(_intra_worker, args=(3, 29539, result), nprocs=3, join=True)
    got = np.load(result)
    ref = oracle_inference(arr, (10, 32, 32), (2, 8, 8),
                           num_output_channels=2, batch_size=3)
    np.testing.assert_allclose(got, ref, rtol=1e-5, atol=1e-6)


# --------------------------------------------------------------------------
# CLI-level worker mode: generate-tasks --task-rank /
# --task-world sharding + the stitch operator, world 2 over gloo — the
# config-3 pipeline end-to-end from the operator surface.
# --------------------------------------------------------------------------
PLUGIN_SRC = '''\
import numpy as np
def execute(chunk):
    # bbox-dependent fill so a misplaced stitch block is visible
    v = (sum(chunk.voxel_offset) % 200) + 1
    return np.full(chunk.shape, v, dtype=np.uint8)
'''


def _cli_pipeline_args(plugin_path, out_npy):
    return ['generate-tasks', '-c', '4', '8', '8',
            '--roi-size', '4', '8', '32',
            'create-chunk',
            'plugin', '-f', plugin_path, '-i', 'chunk', '-o', 'chunk',
            'inference', '-s', '4', '8', '8',
            '--output-patch-overlap', '1', '2', '2',
            '--framework', 'identity', '--num-output-channels', '3',
            '--mask-output-chunk',
            'stitch', '--backend', 'gloo', '-f', out_npy]


@pytest.mark.timeout(240)
def test_cli_worker_mode_world2(tmp_path):
    import subprocess
    import sys
    plugin = str(tmp_path / 'fill_offset.py')
    with open(plugin, 'w') as f:
        f.write(PLUGIN_SRC)

    # single-process run (no env): the truth
    out1 = str(tmp_path / 'vol_w1.npy')
    env = dict(os.environ)
    for k in ('RANK', 'WORLD_SIZE', 'MASTER_ADDR', 'MASTER_PORT',
              'LOCAL_RANK'):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, '-m', 'chunkflow_amd']
        + _cli_pipeline_args(plugin, out1),
        env=env, capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stdout + r.stderr

    # world-2 torchrun-style run: same pipeline, env-driven sharding
    out2 = str(tmp_path / 'vol_w2.npy')
    procs = []
    for rank in range(2):
        env2 = dict(env, RANK=str(rank), WORLD_SIZE='2',
                    MASTER_ADDR='127.0.0.1', MASTER_PORT='29541',
                    LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, '-m', 'chunkflow_amd']
            + _cli_pipeline_args(plugin, out2),
            env=env2, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            text=True))
    for p in procs:
        out, _ = p.communicate(timeout=180)
        assert p.returncode == 0, out

    v1 = np.load(out1)
    v2 = np.load(out2)
    assert v1.shape == (3, 4, 8, 32)
    np.testing.assert_array_equal(v1, v2)
    # content is bbox-dependent: block i along x holds ((8i)%200+1)/255
    for i in range(4):
        np.testing.assert_allclose(
            v1[:, :, :, i * 8:(i + 1) * 8],
            np.float32((8 * i % 200 + 1) / 255.0), rtol=1e-5)


def _worker8(rank, world, port, result_path):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import (init_distributed, shard_tasks,
                                        stitch_to_rank0)
    init_distributed(backend='gloo')
    bboxes = BoundingBoxes.from_manual_setup(
        (4, 8, 8), roi_size=(4, 8, 8 * 8))  # 8 tasks along x
    assert len(bboxes) == 8
    local = {i: torch.full((3, 4, 8, 8), float(i + 1), dtype=torch.float32)
             for i in shard_tasks(list(range(8)), rank, world)}
    vol = stitch_to_rank0(bboxes, local, 3, rank, world, 'cpu')
    if rank == 0:
        np.save(result_path, vol.numpy())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_shard_and_stitch_world8(tmp_path):
    """World 8 — the full 8-GPU-node fan-in shape: rank 0 receives from 7
    concurrent senders."""
    result = str(tmp_path / 'vol8.npy')
    mp.spawn(_worker8, args=(8, 29549, result), nprocs=8, join=True)
    vol = np.load(result)
    assert vol.shape == (3, 4, 8, 64)
    for i in range(8):
        np.testing.assert_array_equal(vol[:, :, :, i * 8:(i + 1) * 8],
                                      np.full((3, 4, 8, 8), i + 1,
                                              dtype=np.float32))


def _empty_rank_worker(rank, world, port, result_path):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import (init_distributed, shard_tasks,
                                        stitch_to_rank0)
    init_distributed(backend='gloo')
    bboxes = BoundingBoxes.from_manual_setup(
        (4, 8, 8), roi_size=(4, 8, 16))  # 2 tasks, 3 ranks: rank 2 empty
    local = {i: torch.full((3, 4, 8, 8), float(i + 1))
             for i in shard_tasks(list(range(2)), rank, world)}
    vol = stitch_to_rank0(bboxes, local, 3, rank, world, 'cpu')
    if rank == 0:
        np.save(result_path, vol.numpy())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_stitch_with_empty_rank(tmp_path):
    """More ranks than tasks: the taskless rank must not deadlock the
    gather (rank 0 posts no receives from it)."""
    result = str(tmp_path / 'vol_empty.npy')
    mp.spawn(_empty_rank_worker, args=(3, 29581, result), nprocs=3,
             join=True)
    vol = np.load(result)
    assert vol.shape == (3, 4, 8, 16)
    np.testing.assert_array_equal(vol[:, :, :, :8], 1.0)
    np.testing.assert_array_equal(vol[:, :, :, 8:], 2.0)
