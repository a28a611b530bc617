"""Phase ablation of the bf16 zring: time modes 0-3 (full / no-mainloop /
no-epilogue / no-staging) to locate the stall (CFX_BF16_MODE)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from chunkflow_amd.fastconv import get_cfx  # noqa: E402

C, N, D, H, W = 28, 24, 32, 256, 256
cl = torch.channels_last_3d
x = torch.randn(N, C, D, H, W, device='cuda').to(torch.bfloat16) \
    .contiguous(memory_format=cl)
w = torch.randn(27, 32, 32).to(torch.bfloat16).cuda().contiguous()
out = torch.empty_like(x)
bias = torch.randn(C, device='cuda') * 0.1
cfx = get_cfx(0)
flops = 2.0 * 27 * C * C * N * D * H * W
USE_BIAS = '--bias' in sys.argv
USE_RES = '--res' in sys.argv
USE_ELU = '--elu' in sys.argv

def t(iters=10):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        cfx.conv3_ndhwc_bf16(x.data_ptr(), w.data_ptr(),
                             bias.data_ptr() if USE_BIAS else None,
                             x.data_ptr() if USE_RES else None,
                             out.data_ptr(), N, D, H, W, C, C,
                             do_elu=USE_ELU)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for _ in range(3):
    t(3)
mode = os.environ.get('CFX_BF16_MODE', '0')
ms = t() * 1e3
print({'mode': mode, 'bias': USE_BIAS, 'res': USE_RES, 'elu': USE_ELU,
       'ms': round(ms, 3), 'TF': round(flops / ms / 1e9, 1)},
      flush=True)
