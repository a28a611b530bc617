#!/usr/bin/env python
"""Benchmark: the `chunkflow inference` hot path on MI355X.

Workload (BASELINE.json configs[1], the headline single-GPU configuration):
one step = full inference of one 512^3 uint8 synthetic chunk through the
3-channel affinity RSUNet (random seeded weights, examples/nets/rsunet.py),
patch 20x256x256, overlap 4x64x64, batch 12, mask_output_chunk=True, f32.
The input chunk is resident in HBM before the timed region; the output stays
in HBM (the PCIe-inclusive rate is reported in DESIGN.md, never as `value`).

--gpus N (launched by torch.distributed.run, one rank per GPU over
RCCL/xGMI): each rank processes its own independent 512^3 chunk per step
(weak scaling — the reference's task-parallel model) and the
disjoint outputs are gathered to rank 0 (BASELINE config 3) inside the timed
region.

Prints ONE JSON line from rank 0 with the whole-job aggregate voxels/sec,
the blend-kernel roofline (HIP-event timing from the C-ABI profiler), and a
bounded-sample CPU baseline (the oracle restatement on this box's host
cores).
"""
import argparse
import json
import os
import sys
import time

os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')

import numpy as np
import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

from chunkflow_amd.cartesian import BoundingBoxes
from chunkflow_amd.chunk import Chunk
from chunkflow_amd.dispatch import init_distributed, stitch_to_rank0
from chunkflow_amd.inferencer import Inferencer

MODEL_FILE = os.path.join(REPO, 'examples', 'nets', 'rsunet.py')
HBM_PEAK_BYTES_PER_S = 8.0e12  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=3)
    p.add_argument('--warmup', type=int, default=1)
    p.add_argument('--chunk-size', type=int, nargs=3,
                   default=(512, 512, 512))
    p.add_argument('--batch-size', type=int, default=12)
    p.add_argument('--patch-size', type=int, nargs=3, default=(20, 256, 256))
    p.add_argument('--overlap', type=int, nargs=3, default=(4, 64, 64))
    p.add_argument('--engine', choices=['rsunet', 'identity'],
                   default='rsunet',
                   help='identity isolates the tiler/blend path')
    p.add_argument('--dtype', choices=['float32', 'bfloat16'],
                   default='float32')
    p.add_argument('--no-cpu-baseline', action='store_true')
    p.add_argument('--cpu-baseline-seconds', type=float, default=15.0)
    p.add_argument('--cpu-baseline-full', action='store_true',
                   help='run the CPU baseline over ALL patches (no '
                        'extrapolation; ~5-6 min)')
    return p.parse_args()


def cpu_baseline(args, chunk_u8: np.ndarray, budget_s: float,
                 full: bool = False) -> dict:
    """The oracle (CPU restatement of the reference path, torch-CPU conv,
    batch 1 like the reference pytorch engine) timed on a bounded sample of
    the same workload; voxels/sec extrapolated from the patch fraction."""
    from oracle.inference import patch_slices_list, blend_into
    from oracle.patch_mask import make_patch_mask
    from chunkflow_amd.model_loader import load_source

    # 3-D conv on torch-CPU stops scaling (and regresses) past ~32 threads;
    # use the best setting for the box rather than oversubscribing
    cores = min(32, os.cpu_count())
    torch.set_num_threads(cores)
    ps, ov = tuple(args.patch_size), tuple(args.overlap)
    mask = make_patch_mask(ps, ov)
    slices = patch_slices_list(chunk_u8.shape, ps, ov)
    n_total = len(slices)
    chunk_f32 = chunk_u8.astype(np.float32)
    chunk_f32 /= 255.0
    out = np.zeros((3,) + chunk_u8.shape, dtype=np.float32)

    if args.engine == 'rsunet':
        model = load_source(MODEL_FILE).InstantiatedModel
        model.eval()

        def forward(patch5d):
            with torch.no_grad():
                return model(torch.from_numpy(patch5d)).numpy()
    else:
        def forward(patch5d):
            return np.repeat(patch5d, 3, axis=1)

    done = 0
    t0 = time.perf_counter()
    for (iz, iy, ix), (oz, oy, ox) in slices:
        buf = chunk_f32[iz:iz + ps[0], iy:iy + ps[1],
                        ix:ix + ps[2]][None, None]
        res = forward(np.ascontiguousarray(buf)) * mask
        blend_into(out, (0, 0, 0), res[0], (oz, oy, ox))
        done += 1
        if not full and time.perf_counter() - t0 > budget_s and done >= 2:
            break
    elapsed = time.perf_counter() - t0
    vox = chunk_u8.size * (done / n_total)
    return {
        'value': vox / elapsed,
        'unit': 'voxels/s',
        'cores': cores,
        'kind': 'port',
        'sample': f'{done}/{n_total} patches of the 512^3 config-2 workload,'
                  f' oracle numpy tiler/blend + torch-CPU '
                  f'{args.engine} at batch 1, {elapsed:.1f}s'
                  + (' (FULL RUN, no extrapolation)' if full else ''),
    }


def read_pmc_traffic(args):
    """(bytes, provenance) of the blend kernel's per-launch HBM traffic
    from a committed rocprofv3 PMC measurement (collected offline —
    rocprofv3 PMC passes cannot run inside the timed bench). The value is
    only attached when the run's workload MATCHES the mix it was measured
    on (per-launch blend sizes differ between configs); otherwise
    (None, reason)."""
    if args.dtype == 'float32' and tuple(args.patch_size) == (20, 256, 256):
        path = os.path.join(REPO, 'profiles', 'pmc_traffic.json')
    elif args.dtype == 'bfloat16' and             tuple(args.patch_size) == (32, 256, 256):
        path = os.path.join(REPO, 'profiles', 'pmc_traffic_bf16.json')
    else:
        return None, 'no PMC measurement for this workload mix'
    if not os.path.exists(path):
        return None, 'no PMC measurement for this workload mix'
    try:
        with open(path) as f:
            d = json.load(f)
        return (float(d['blend_bytes_per_launch']),
                d.get('provenance', path))
    except Exception:
        return None, None


def main():
    args = parse_args()
    rank, world = init_distributed()
    if world > 1:
        assert world == args.gpus, (world, args.gpus)
    local_rank = int(os.environ.get('LOCAL_RANK', 0))
    if not torch.cuda.is_available():
        raise RuntimeError('bench.py requires a GPU (the CPU plumbing path '
                           'is not the product path)')
    # modulo keeps the 1:1 mapping on a full node and lets the gloo
    # dress-rehearsal run several ranks on a single-GPU box
    local_rank = local_rank % torch.cuda.device_count()
    torch.cuda.set_device(local_rank)
    device = f'cuda:{local_rank}'
    torch.manual_seed(0)
    np.random.seed(0)

    cz, cy, cx = args.chunk_size
    bboxes = BoundingBoxes.from_manual_setup(
        (cz, cy, cx), roi_size=(cz, cy, cx * world))
    assert len(bboxes) == world
    my_bbox = bboxes[rank]

    # synthetic input (deterministic sin pattern, chunk/base.py:170-179),
    # uploaded to HBM BEFORE the timed region
    host_chunk = Chunk.create(size=(cz, cy, cx), dtype='uint8',
                              pattern='sin', voxel_offset=my_bbox.start)
    dev_chunk = host_chunk.to_device(device)

    if args.engine == 'rsunet':
        inferencer = Inferencer(
            MODEL_FILE, None, args.patch_size,
            output_patch_overlap=args.overlap, framework='pytorch',
            num_output_channels=3, batch_size=args.batch_size,
            mask_output_chunk=True,
            dtype='bfloat16' if args.dtype == 'bfloat16' else 'float32',
            compute_device=device)
    else:
        inferencer = Inferencer(
            None, None, args.patch_size, output_patch_overlap=args.overlap,
            framework='identity', num_output_channels=3,
            batch_size=args.batch_size, mask_output_chunk=True,
            compute_device=device)
    assert inferencer.ops.is_hip

    def step():
        out = inferencer(dev_chunk)
        if world > 1:
            t = out.array
            if t.dtype != torch.float32:
                t = t.to(torch.float32)
            # one task per rank (task i belongs to rank i % world == i)
            stitch_to_rank0(bboxes, {rank: t}, 3, rank, world, device)
        return out

    import torch.distributed as dist

    def barrier_sync():
        if world > 1:
            dist.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()

    cfx = inferencer.ops.cfx
    cfx.profile_reset()
    cfx.profile_enable(True)
    # the fastconv kernels run on their own context; profile it too
    conv_ctx = None
    if args.engine == 'rsunet':
        try:
            from chunkflow_amd.fastconv import _CTX
            conv_ctx = _CTX.get(local_rank)
        except Exception:
            conv_ctx = None
    if conv_ctx is not None:
        conv_ctx.profile_reset()
        conv_ctx.profile_enable(True)

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if world > 1:
        # gloo (the CFX_DIST_BACKEND dress-rehearsal path) moves CPU
        # tensors only; nccl/RCCL wants the device tensor
        red_dev = 'cpu' if dist.get_backend() == 'gloo' else device
        t = torch.tensor([elapsed], device=red_dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    blend = cfx.profile_get('blend')
    # the hand MFMA conv kernels (fastconv) publish FLOPs in the bytes slot
    conv = None
    conv_stream = None
    if conv_ctx is not None:
        conv = conv_ctx.profile_get('conv')
        conv_stream = conv_ctx.profile_get('conv_stream')
        conv_ctx.profile_enable(False)
    cfx.profile_enable(False)

    if rank != 0:
        if world > 1:
            dist.barrier()  # let rank 0 finish the CPU baseline + print
        return

    voxels = float(cz) * cy * cx * world * args.steps
    value = voxels / elapsed

    traffic, traffic_src = read_pmc_traffic(args)
    conv_roofline = None
    if conv and conv['count'] > 0 and conv['total_ms'] > 0:
        tflops = conv['bytes'] / (conv['total_ms'] / 1e3) / 1e12
        # useful FLOP rate of the MFMA ring convs alone (the HBM-stream
        # convs report separately below); peak = the dtype's dense MFMA
        # peak (MI355X_MICROARCH.md)
        peak = 157.3 if args.dtype == 'float32' else 2500.0
        conv_roofline = {
            'bound': 'mfma', 'achieved_tflops': tflops,
            'peak_tflops': peak,
            'frac': tflops / peak, 'launches': conv['count'],
            'kernel': 'MFMA z-ring convs (fastconv), ' + args.dtype,
        }
        if conv_stream and conv_stream['count'] > 0 \
                and conv_stream['total_ms'] > 0:
            conv_roofline['stream_convs'] = {
                'bound': 'hbm',
                'achieved_tflops': conv_stream['bytes']
                / (conv_stream['total_ms'] / 1e3) / 1e12,
                'launches': conv_stream['count'],
                'kernel': 'upconv/conv155 HBM-stream convs',
            }
    roofline = None
    if blend['count'] > 0 and blend['total_ms'] > 0:
        achieved = blend['bytes'] / (blend['total_ms'] / 1e3)
        roofline = {
            'bound': 'hbm',
            'achieved': achieved / 1e9,
            'peak': HBM_PEAK_BYTES_PER_S / 1e9,
            'unit': 'GB/s',
            'frac': achieved / HBM_PEAK_BYTES_PER_S,
            'traffic': traffic,
            'traffic_source': traffic_src,
            'kernel': 'k_blend',
            'launches': blend['count'],
            'algorithmic_bytes_per_launch': blend['bytes'] / blend['count'],
            'avg_launch_ms': blend['total_ms'] / blend['count'],
            'conv': conv_roofline,
        }

    cpu = None
    if not args.no_cpu_baseline and world == 1:
        cpu = cpu_baseline(args, host_chunk.array,
                           args.cpu_baseline_seconds,
                           full=args.cpu_baseline_full)

    result = {
        'metric': 'output_voxels_per_sec',
        'value': value,
        'unit': 'voxels/s',
        'n_gpus': world,
        'steps': args.steps,
        'warmup': args.warmup,
        'ms_per_step': elapsed / args.steps * 1e3,
        'higher_is_better': True,
        'scaling': 'weak',
        'vs_baseline': None,  # the reference publishes no number
        'dtype': 'f32' if args.dtype == 'float32' else 'bf16',
        'data': 'synthetic',
        'config': {
            'workload': 'config2-512cube-rsunet-affinity'
            if args.engine == 'rsunet' else 'identity-tiler-blend-only',
            'chunk': list(args.chunk_size),
            'patch': list(args.patch_size),
            'overlap': list(args.overlap),
            'batch_size': args.batch_size,
            'engine': args.engine,
            'chunks_per_step': world,
            'stitch': (('rccl' if dist.get_backend() == 'nccl'
                        else dist.get_backend())
                       + '-p2p-gather-to-rank0') if world > 1 else None,
        },
        'roofline': roofline,
        'cpu_baseline': cpu,
    }
    print(json.dumps(result))
    if world > 1:
        dist.barrier()


if __name__ == '__main__':
    main()
