"""A/B the round-2 kernels vs torch/MIOpen on the RSUNet shapes:
up/down-sampling convs (csrc/updown.hip) and the sliced bf16 ring
(C=36/48). Drained-queue timing (tools/conv_probe.py lesson). Writes JSON
to gpurun_out/updown_probe.json."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn as nn

from chunkflow_amd.fastconv import (CfxUpConv3d, CfxDownConv3d,
                                    CfxConv3dBF16, CfxConvIn155,
                                    CfxConvOut155)

torch.backends.cudnn.benchmark = True
cl = torch.channels_last_3d
RES = {}


def timeit(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def probe_updown(kind, C, K, N, D, H, W, bf16):
    torch.manual_seed(0)
    dt = torch.bfloat16 if bf16 else torch.float32
    if kind == 'up':
        conv = nn.ConvTranspose3d(C, K, (1, 2, 2), stride=(1, 2, 2)).cuda()
        repl = CfxUpConv3d(conv, 0, bf16=bf16).cuda()
    else:
        conv = nn.Conv3d(C, K, (1, 2, 2), stride=(1, 2, 2)).cuda()
        repl = CfxDownConv3d(conv, 0, bf16=bf16).cuda()
    tconv = conv.to(dt).to(memory_format=cl)
    x = torch.randn(N, C, D, H, W, device='cuda').to(dt).contiguous(
        memory_format=cl)
    with torch.no_grad():
        want = tconv(x).float()
        got = repl(x).float()
        err = (got - want).abs().max().item()
        t_my = timeit(lambda: repl(x))
        t_to = timeit(lambda: tconv(x))
    name = f'{kind}_{C}to{K}_{"bf16" if bf16 else "f32"}_{H}x{W}'
    RES[name] = {'mine_ms': t_my * 1e3, 'torch_ms': t_to * 1e3,
                 'speedup': t_to / t_my, 'max_err': err}
    print(name, RES[name], flush=True)


def probe_sliced(C, N, D, H, W):
    torch.manual_seed(0)
    conv = nn.Conv3d(C, C, 3, padding=1).cuda()
    repl = CfxConv3dBF16(conv, 0).cuda()
    tconv = conv.to(torch.bfloat16).to(memory_format=cl)
    x = torch.randn(N, C, D, H, W, device='cuda').to(torch.bfloat16) \
        .contiguous(memory_format=cl)
    with torch.no_grad():
        want = tconv(x).float()
        got = repl._run(x).float()
        err = (got - want).abs().max().item()
        mean_err = (got - want).abs().mean().item()
        t_my = timeit(lambda: repl._run(x))
        t_to = timeit(lambda: tconv(x))
    useful_tf = 2.0 * 27 * C * C * N * D * H * W / 1e12
    RES[f'ring_bf16_C{C}_{H}x{W}'] = {
        'mine_ms': t_my * 1e3, 'torch_ms': t_to * 1e3,
        'speedup': t_to / t_my, 'useful_tf_mine': useful_tf / t_my,
        'useful_tf_torch': useful_tf / t_to,
        'max_err': err, 'mean_err': mean_err}
    print(f'ring_bf16_C{C}', RES[f'ring_bf16_C{C}_{H}x{W}'], flush=True)


def probe_conv155(N, D, H, W, bf16):
    torch.manual_seed(0)
    conv = nn.Conv3d(1, 28, (1, 5, 5), padding=(0, 2, 2)).cuda()
    repl = CfxConvIn155(conv, 0, bf16=bf16).cuda()
    dt = torch.bfloat16 if bf16 else torch.float32
    tconv = conv.to(dt).to(memory_format=cl)
    x = torch.randn(N, 1, D, H, W, device='cuda').to(dt).contiguous(
        memory_format=cl)
    with torch.no_grad():
        want = tconv(x).float()
        got = repl(x).float()
        err = (got - want).abs().max().item()
        t_my = timeit(lambda: repl(x))
        t_to = timeit(lambda: tconv(x))
    name = f'conv155_{"bf16" if bf16 else "f32"}_{N}x{D}x{H}x{W}'
    RES[name] = {'mine_ms': t_my * 1e3, 'torch_ms': t_to * 1e3,
                 'speedup': t_to / t_my, 'max_err': err}
    print(name, RES[name], flush=True)


def probe_conv155_out(N, D, H, W, bf16):
    torch.manual_seed(0)
    conv = nn.Conv3d(28, 3, (1, 5, 5), padding=(0, 2, 2)).cuda()
    repl = CfxConvOut155(conv, 0, bf16=bf16).cuda()
    dt = torch.bfloat16 if bf16 else torch.float32
    tconv = conv.to(dt).to(memory_format=cl)
    x = torch.randn(N, 28, D, H, W, device='cuda').to(dt).contiguous(
        memory_format=cl)
    with torch.no_grad():
        want = tconv(x).float()
        got = repl(x).float()
        err = (got - want).abs().max().item()
        t_my = timeit(lambda: repl(x))
        t_to = timeit(lambda: tconv(x))
    name = f'conv155_out_{"bf16" if bf16 else "f32"}_{N}x{D}x{H}x{W}'
    RES[name] = {'mine_ms': t_my * 1e3, 'torch_ms': t_to * 1e3,
                 'speedup': t_to / t_my, 'max_err': err}
    print(name, RES[name], flush=True)


def main():
    N, D = 12, 20  # config-2-like depth; config-5 uses N=24 D=32
    # RSUNet up/down shapes (H, W = input dims of the op)
    for bf16 in (True, False):
        probe_updown('up', 64, 48, N, D, 32, 32, bf16)
        probe_updown('up', 48, 36, N, D, 64, 64, bf16)
        probe_updown('up', 36, 28, N, D, 128, 128, bf16)
        probe_updown('down', 28, 36, N, D, 256, 256, bf16)
        probe_updown('down', 36, 48, N, D, 128, 128, bf16)
        probe_updown('down', 48, 64, N, D, 64, 64, bf16)
    probe_sliced(36, N, D, 128, 128)
    probe_sliced(48, N, D, 64, 64)
    # config-5 geometry
    probe_sliced(36, 24, 32, 128, 128)
    probe_sliced(48, 24, 32, 64, 64)
    probe_conv155(12, 20, 256, 256, False)
    probe_conv155(24, 32, 256, 256, True)
    probe_conv155_out(12, 20, 256, 256, False)
    probe_conv155_out(24, 32, 256, 256, True)
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/updown_probe.json', 'w') as f:
        json.dump(RES, f, indent=1)


if __name__ == '__main__':
    main()
