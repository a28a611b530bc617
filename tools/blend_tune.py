"""Micro-bench the batched blend kernel variants on realistic groups.

Builds the config-2 512^3 geometry (288 patches, first-fit groups), fills a
3-channel engine-output-sized batch, and times cfx_blend_batch over all
groups, sweeping CFX_BLEND_G / CFX_BLEND_NT via subprocesses (the knobs are
read once per process).

Usage: python tools/blend_tune.py          # run all variants
       python tools/blend_tune.py --one    # run with current env only
"""
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run_one():
    import numpy as np
    import torch
    from chunkflow_amd.ops import HipOps
    from chunkflow_amd.grouping import disjoint_groups
    from chunkflow_amd.patch_mask import make_patch_mask

    size = (512, 512, 512)
    ps, ov = (20, 256, 256), (4, 64, 64)
    C = 3
    ops = HipOps(0)
    # the config-2 tail-clamped patch grid (same loop as the product tiler)
    stride = tuple(p - o for p, o in zip(ps, ov))
    axes = []
    for d in range(3):
        ax = []
        for i in range(0, size[d] - ov[d], stride[d]):
            ax.append(min(i, size[d] - ps[d]))
        axes.append(ax)
    starts = np.array([(z, y, x) for z in axes[0] for y in axes[1]
                       for x in axes[2]], dtype=np.int64)
    groups = disjoint_groups(starts, ps, size)
    mask = torch.from_numpy(make_patch_mask(ps, ov).copy()).cuda()
    out = torch.zeros((C,) + size, dtype=torch.float32, device='cuda')
    # a fat patch buffer: reuse 32 slots round-robin as the "engine output"
    patch = torch.rand((32, C) + ps, dtype=torch.float32, device='cuda')
    patch *= 0.001  # keep sums < 1 irrelevant here

    items_per_group = []
    total_bytes = 0.0
    for idx in groups:
        items = np.concatenate(
            [(idx % 32)[:, None].astype(np.int32),
             starts[idx].astype(np.int32)], axis=1)
        items_per_group.append(items)
        for s in starts[idx]:
            lo = np.maximum(s, 0)
            hi = np.minimum(s + np.array(ps), np.array(size))
            rv = float(np.prod(np.maximum(hi - lo, 0)))
            total_bytes += rv * (C * 12 + 4)

    def sweep():
        for items in items_per_group:
            ops.blend_batch(out, patch, items, mask=mask)

    for _ in range(3):
        sweep()
    torch.cuda.synchronize()
    reps = 10
    t0 = time.perf_counter()
    for _ in range(reps):
        sweep()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    print(json.dumps({
        'G': os.environ.get('CFX_BLEND_G', '4'),
        'NT': os.environ.get('CFX_BLEND_NT', '0'),
        'groups': len(groups),
        'ms': dt * 1e3,
        'GBps': total_bytes / dt / 1e9,
        'frac_of_8TBps': total_bytes / dt / 8e12,
    }))


def main():
    if '--one' in sys.argv:
        run_one()
        return
    for g in ('2', '4', '8'):
        for nt in ('0', '1'):
            env = dict(os.environ, CFX_BLEND_G=g, CFX_BLEND_NT=nt)
            subprocess.run([sys.executable, __file__, '--one'], env=env,
                           check=False)


if __name__ == '__main__':
    main()
