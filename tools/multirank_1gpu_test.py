"""Config-3 placement proof on hardware with the means a 1-GPU box allows:
TWO processes, each running the real HIP inference of its own task shard on
cuda:0 (RCCL refuses two ranks on one device — profiles/rccl_hw_r02.json),
stitched to rank 0 over gloo with the SAME dispatch.stitch_to_rank0 code
bench.py times under RCCL. Rank 0 checks bit-exact placement of every
rank's sub-volume against a single-process reference run.

Writes gpurun_out/multirank_1gpu.json.
"""
import hashlib
import json
import os
import sys

import numpy as np
import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

CZ, CY, CX = 128, 256, 256
PATCH = (20, 128, 128)
OV = (4, 32, 32)


def _run_chunk(seed):
    from chunkflow_amd.chunk import Chunk
    from chunkflow_amd.inferencer import Inferencer
    rng = np.random.RandomState(seed)
    arr = rng.randint(0, 256, size=(CZ, CY, CX)).astype(np.uint8)
    inf = Inferencer(None, None, PATCH, output_patch_overlap=OV,
                     framework='identity', num_output_channels=3,
                     batch_size=4, mask_output_chunk=True,
                     compute_device='cuda:0')
    out = inf(Chunk(arr))
    t = out.array
    return t.float().cpu()


def _worker(rank, world, port, outdir):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import init_distributed, stitch_to_rank0
    init_distributed(backend='gloo')
    bboxes = BoundingBoxes.from_manual_setup(
        (CZ, CY, CX), roi_size=(CZ, CY, CX * world))
    local = {i: _run_chunk(seed=100 + i)
             for i in range(world) if i % world == rank}
    vol = stitch_to_rank0(bboxes, local, 3, rank, world, 'cpu')
    if rank == 0:
        np.save(os.path.join(outdir, 'stitched.npy'), vol.numpy())
    dist.barrier()
    dist.destroy_process_group()


def main():
    os.makedirs('gpurun_out', exist_ok=True)
    outdir = 'gpurun_out'
    mp.spawn(_worker, args=(2, 29571, outdir), nprocs=2, join=True)
    vol = np.load(os.path.join(outdir, 'stitched.npy'))
    # single-process reference: the same two chunks, placed manually
    ref = np.concatenate([_run_chunk(100).numpy(), _run_chunk(101).numpy()],
                         axis=3)
    exact = bool((vol == ref).all())
    res = {
        'world': 2, 'compute_device': 'cuda:0 (both ranks)',
        'stitch': 'dispatch.stitch_to_rank0 over gloo (RCCL refuses '
                  'same-device world-2; placement logic identical)',
        'volume_shape': list(vol.shape),
        'bit_exact_placement': exact,
        'sha256_stitched': hashlib.sha256(vol.tobytes()).hexdigest()[:16],
        'sha256_reference': hashlib.sha256(ref.tobytes()).hexdigest()[:16],
    }
    with open('gpurun_out/multirank_1gpu.json', 'w') as f:
        json.dump(res, f, indent=1)
    print(res)
    assert exact
    os.remove(os.path.join(outdir, 'stitched.npy'))


if __name__ == '__main__':
    main()
