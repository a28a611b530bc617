"""Measure the H2D/D2H edges of one config-2 chunk (for DESIGN.md's
PCIe-inclusive note; never part of `value`)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

u8 = np.random.randint(0, 256, size=(512, 512, 512), dtype=np.uint8)
t0 = time.perf_counter()
d = torch.from_numpy(u8).cuda()
torch.cuda.synchronize()
h2d = time.perf_counter() - t0
out = torch.rand(3, 512, 512, 512, device='cuda')
torch.cuda.synchronize()
t0 = time.perf_counter()
h = out.cpu()
torch.cuda.synchronize()
d2h = time.perf_counter() - t0
print({'h2d_u8_s': h2d, 'h2d_GBps': u8.nbytes / h2d / 1e9,
       'd2h_f32_s': d2h, 'd2h_GBps': out.numel() * 4 / d2h / 1e9})
