"""Time the RSUNet batch forward under one MIOpen/layout setting.

Usage (one setting per process — MIOpen find mode is fixed at first conv):
    MIOPEN_FIND_MODE=... python tools/conv_sweep.py [--channels-last]
        [--batch N] [--bf16] [--benchmark]
Prints one JSON line: {setting..., ms_per_batch, tflops}.
"""
import argparse
import json
import os
import sys
import time

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from chunkflow_amd.model_loader import load_source


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--batch', type=int, default=12)
    p.add_argument('--channels-last', action='store_true')
    p.add_argument('--bf16', action='store_true')
    p.add_argument('--benchmark', action='store_true')
    p.add_argument('--iters', type=int, default=4)
    p.add_argument('--patch', type=int, nargs=3, default=(20, 256, 256))
    args = p.parse_args()

    torch.backends.cudnn.benchmark = args.benchmark
    model = load_source(os.path.join(
        REPO, 'examples', 'nets', 'rsunet.py')).InstantiatedModel
    model = model.cuda().eval()
    x = torch.rand(args.batch, 1, *args.patch, device='cuda')
    if args.bf16:
        model = model.to(torch.bfloat16)
        x = x.to(torch.bfloat16)
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last_3d)
        x = x.contiguous(memory_format=torch.channels_last_3d)

    with torch.no_grad():
        for _ in range(2):
            model(x)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            model(x)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / args.iters * 1e3

    print(json.dumps({
        'find_mode': os.environ.get('MIOPEN_FIND_MODE', '<default>'),
        'benchmark': args.benchmark,
        'channels_last': args.channels_last,
        'batch': args.batch,
        'bf16': args.bf16,
        'ms_per_batch': ms,
        'patches_per_s': args.batch / (ms / 1e3),
    }))


if __name__ == '__main__':
    main()
