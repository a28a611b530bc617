"""Per-phase s_memtime breakdown of the bf16 ring (CFX_ZRING_PL=12):
launches the timing clone (results intentionally wrong) and reads the
per-workgroup cycle sums it dumps into the output buffer."""
import os
import sys

os.environ['CFX_ZRING_PL'] = '12'
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from chunkflow_amd.fastconv import get_cfx

N, C, D, H, W = 24, 28, 32, 256, 256
cl = torch.channels_last_3d
x = torch.randn(N, C, D, H, W, device='cuda').to(torch.bfloat16) \
    .contiguous(memory_format=cl)
w = torch.randn(27, 32, 32).to(torch.bfloat16).cuda().contiguous()
out = torch.zeros(N * C * D * H * W, dtype=torch.bfloat16,
                  device='cuda')  # flat raw buffer (kernel sees a pointer)
cfx = get_cfx(0)
for _ in range(3):  # warm
    cfx.conv3_ndhwc_bf16(x.data_ptr(), w.data_ptr(), None, None,
                         out.data_ptr(), N, D, H, W, C, C)
torch.cuda.synchronize()
cfx.conv3_ndhwc_bf16(x.data_ptr(), w.data_ptr(), None, None,
                     out.data_ptr(), N, D, H, W, C, C)
torch.cuda.synchronize()
n_wg = (W // 32) * (H // 8) * N
rec = out[:n_wg * 6 * 4].view(torch.int64).view(n_wg, 6).cpu()
rec = rec.double()
names = ['plane_load_issue', 'mfma_phase1(36)', 'store+barrier',
         'mfma_phase2(18)', 'epilogue+rotate', 'total']
mean = rec.mean(dim=0) / D  # per z
print(f'{n_wg} WGs, per-z cycles (wave-0 view, D={D}):')
tot = float(mean[5])
for i, n in enumerate(names):
    print(f'  {n:22s} {float(mean[i]):9.0f} cyc  {float(mean[i])/tot*100:5.1f}%')
print('MFMA-issue floor per z per wave-pair: 54*2*32 =', 54*2*32, 'cyc')
