"""A/B the hand-written MFMA conv vs torch/MIOpen per RSUNet shape, and
check numerics vs a float64 CPU reference."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from chunkflow_amd.fastconv import get_cfx

torch.backends.cudnn.benchmark = True
cl = torch.channels_last_3d

def probe(C, D, H, W, N=12, iters=10, w32=False, zring=False):
    torch.manual_seed(0)
    x = torch.randn(N, C, D, H, W, device='cuda').contiguous(memory_format=cl)
    wt = torch.randn(C, C, 3, 3, 3, device='cuda') * (1.0 / (27 * C) ** 0.5)
    bias = torch.randn(C, device='cuda') * 0.1
    wtap = wt.permute(2, 3, 4, 1, 0).reshape(27, C, C).contiguous()
    out = torch.empty_like(x)
    cfx = get_cfx(0)

    # numerics vs fp64 CPU
    ref = F.conv3d(x.double().cpu(), wt.double().cpu(), bias.double().cpu(),
                   padding=1)[:2]
    cfx.conv3_ndhwc(x.data_ptr(), wtap.data_ptr(), bias.data_ptr(), None,
                    out.data_ptr(), N, D, H, W, C, C, w32=w32, zring=zring)
    torch.cuda.synchronize()
    err = (out[:2].double().cpu() - ref).abs().max().item()
    scale = ref.abs().max().item()

    def t_mine():
        t0 = time.perf_counter()
        for _ in range(iters):
            cfx.conv3_ndhwc(x.data_ptr(), wtap.data_ptr(), bias.data_ptr(),
                            None, out.data_ptr(), N, D, H, W, C, C, w32=w32, zring=zring)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    wm = wt.contiguous(memory_format=cl)
    def t_torch():
        t0 = time.perf_counter()
        for _ in range(iters):
            F.conv3d(x, wm, bias, padding=1)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    for _ in range(3):
        t_mine(); t_torch()
    tm, tt = t_mine(), t_torch()
    flops = 2.0 * 27 * C * C * N * D * H * W
    print({'C': C, 'w32': w32, 'zring': zring, 'shape': (N, D, H, W), 'err_vs_fp64': err,
           'rel': err / scale,
           'mine_ms': tm * 1e3, 'torch_ms': tt * 1e3,
           'mine_TF': flops / tm / 1e12, 'torch_TF': flops / tt / 1e12,
           'speedup': tt / tm}, flush=True)

if __name__ == '__main__' and '--bf16' not in sys.argv:
    probe(28, 20, 256, 256)
    probe(28, 20, 256, 256, w32=True)
    probe(28, 20, 256, 256, zring=True)
    probe(36, 20, 128, 128, zring=True)
    probe(36, 20, 128, 128)
    probe(48, 20, 64, 64)
    probe(64, 20, 32, 32)


def probe_bf16(C=28, D=20, H=256, W=256, N=24, iters=8):
    from chunkflow_amd.fastconv import CfxConv3dBF16
    torch.manual_seed(0)
    conv = torch.nn.Conv3d(C, C, 3, padding=1).cuda()
    m = CfxConv3dBF16(conv).cuda()
    x = (torch.randn(N, C, D, H, W, device='cuda') * 0.3) \
        .to(torch.bfloat16).contiguous(memory_format=cl)
    convb = conv.to(torch.bfloat16).to(memory_format=cl)
    got = m(x).float()
    ref = torch.nn.functional.conv3d(x, convb.weight, convb.bias,
                                     padding=1).float()
    err = (got - ref).abs().max().item()

    def t(f):
        torch.cuda.synchronize()  # drain the queue: un-synced warmup work
        t0 = time.perf_counter()  # was inflating 'mine' by ~7.5 ms/iter
        for _ in range(iters):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    for _ in range(3):
        m(x); torch.nn.functional.conv3d(x, convb.weight, convb.bias,
                                         padding=1)
    torch.cuda.synchronize()
    tm = t(lambda: m(x))
    tt = t(lambda: torch.nn.functional.conv3d(x, convb.weight, convb.bias,
                                              padding=1))
    flops = 2.0 * 27 * C * C * N * D * H * W
    print({'bf16': True, 'C': C, 'shape': (N, D, H, W),
           'err_vs_torch_bf16': err, 'mine_ms': tm * 1e3,
           'torch_ms': tt * 1e3, 'mine_TF': flops / tm / 1e12,
           'torch_TF': flops / tt / 1e12, 'speedup': tt / tm}, flush=True)


if __name__ == '__main__' and '--bf16' in sys.argv:
    probe_bf16(28, 32, 256, 256, N=24)
