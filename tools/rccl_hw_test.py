"""Exercise the RCCL stitch path on real MI355X hardware.

The builder-side gpurun pool leases 1-GPU boxes, so true multi-GPU RCCL
cannot run here (the driver's round-end 8-GPU bench covers that). This
tool extracts what CAN be proven on one GPU:

  1. world=2 over RCCL with both ranks on the same device — expected to
     be refused by NCCL/RCCL (duplicate GPU); we record the exact error.
  2. world=1 over RCCL: init_process_group('nccl'), a device all_reduce,
     and the full dispatch.stitch_to_rank0 code path (degenerate world) —
     proves RCCL initializes and executes collectives on this hardware
     and that the stitch code runs on-device end to end.

Writes gpurun_out/rccl_hw.json.
"""
import json
import os
import sys
import traceback

import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

RES = {}


def _dup_worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    try:
        torch.cuda.set_device(0)
        dist.init_process_group('nccl')
        t = torch.ones(8, device='cuda:0')
        dist.all_reduce(t)
        torch.cuda.synchronize()
        q.put((rank, 'ok', float(t[0].item())))
        dist.destroy_process_group()
    except Exception as e:
        q.put((rank, 'error', f'{type(e).__name__}: {e}'[:500]))


def try_world2_same_gpu():
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dup_worker, args=(r, 2, 29561, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = []
    for p in procs:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
            outs.append(('?', 'timeout', ''))
    while not q.empty():
        outs.append(q.get())
    RES['world2_same_gpu'] = [list(o) for o in outs]
    print('world2_same_gpu:', outs, flush=True)


def world1_rccl():
    import torch.distributed as dist
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import stitch_to_rank0
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT='29562',
                      RANK='0', WORLD_SIZE='1')
    try:
        dist.init_process_group('nccl', rank=0, world_size=1)
        t = torch.arange(16, dtype=torch.float32, device='cuda:0')
        dist.all_reduce(t)
        dist.barrier()
        torch.cuda.synchronize()
        bboxes = BoundingBoxes.from_manual_setup((4, 8, 8),
                                                 roi_size=(4, 8, 16))
        local = {i: torch.full((3, 4, 8, 8), float(i + 1),
                               device='cuda:0')
                 for i in range(2)}
        vol = stitch_to_rank0(bboxes, local, 3, 0, 1, 'cuda:0')
        ok = (vol[:, :, :, :8] == 1).all() and (vol[:, :, :, 8:] == 2).all()
        RES['world1_rccl'] = {
            'init': 'ok', 'all_reduce_sum0': float(t[0].item()),
            'stitch_placement_ok': bool(ok),
            'backend': dist.get_backend(),
            'nccl_version': list(torch.cuda.nccl.version()),
        }
        dist.destroy_process_group()
    except Exception as e:
        RES['world1_rccl'] = {'error': traceback.format_exc()[-500:]}
    print('world1_rccl:', RES['world1_rccl'], flush=True)


if __name__ == '__main__':
    try_world2_same_gpu()
    world1_rccl()
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/rccl_hw.json', 'w') as f:
        json.dump(RES, f, indent=1)
