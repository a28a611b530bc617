"""Phase ablation of the f32 zring (CFX_F32_MODE: 0 full, 1 no-mainloop,
3 no-staging) at the config-2 dominant shape."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from chunkflow_amd.fastconv import get_cfx  # noqa: E402

C, N, D, H, W = 28, 12, 20, 256, 256
cl = torch.channels_last_3d
x = torch.randn(N, C, D, H, W, device='cuda').contiguous(memory_format=cl)
w = torch.randn(27, C, C, device='cuda').contiguous()
out = torch.empty_like(x)
cfx = get_cfx(0)
flops = 2.0 * 27 * C * C * N * D * H * W

def t(iters=10):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        cfx.conv3_ndhwc(x.data_ptr(), w.data_ptr(), None, None,
                        out.data_ptr(), N, D, H, W, C, C, zring=True)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for _ in range(3):
    t(3)
ms = t() * 1e3
print({'mode': os.environ.get('CFX_F32_MODE', '0'), 'ms': round(ms, 3),
       'TF': round(flops / ms / 1e9, 1)}, flush=True)
