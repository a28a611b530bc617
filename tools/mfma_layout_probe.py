"""Empirically verify the v_mfma_f32_32x32x16_bf16 fragment layout.

Hypothesis (CDNA pattern): A[i][k]: lane l holds i = l&31, k = 8*(l>>5)+e
for e = 0..7 (bf16 pairs k=2j,2j+1 per VGPR); B[k][j]: j = l&31,
k = 8*(l>>5)+e; D (f32x16): col j = l&31, row = (reg&3) + 8*(reg>>2) +
4*(l>>5). Asymmetric A and B catch any transpose (guide G9).
"""
import os
import subprocess

import numpy as np

SRC = r'''
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
typedef __bf16 bf16;
typedef bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

extern "C" __global__ void probe16(const bf16* A /*16x32 row-major*/,
                                   const bf16* B /*32x16 row-major*/,
                                   float* D /*16x16 row-major*/) {
    int l = threadIdx.x;
    bf16x8 a, b;
    typedef float f32x4 __attribute__((ext_vector_type(4)));
    for (int e = 0; e < 8; ++e) {
        int k = 8 * (l >> 4) + e;
        a[e] = A[(l & 15) * 32 + k];      // A[i][k]
        b[e] = B[k * 16 + (l & 15)];      // B[k][j]
    }
    f32x4 acc = {};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    for (int r = 0; r < 4; ++r) {
        int row = (l >> 4) * 4 + r;       // C/D: col=lane&15, row=(l>>4)*4+r
        D[row * 16 + (l & 15)] = acc[r];
    }
}

extern "C" __global__ void probe(const bf16* A /*32x16 row-major*/,
                                 const bf16* B /*16x32 row-major*/,
                                 float* D /*32x32 row-major*/) {
    int l = threadIdx.x;
    bf16x8 a, b;
    for (int e = 0; e < 8; ++e) {
        int k = 8 * (l >> 5) + e;
        a[e] = A[(l & 31) * 16 + k];      // A[i][k]
        b[e] = B[k * 32 + (l & 31)];      // B[k][j]
    }
    f32x16 acc = {};
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    for (int r = 0; r < 16; ++r) {
        int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
        D[row * 32 + (l & 31)] = acc[r];
    }
}
'''


def main():
    import torch
    here = '/tmp/mfma_probe'
    os.makedirs(here, exist_ok=True)
    import ctypes
    rng = np.random.RandomState(0)
    A = (rng.randn(32, 16) * 0.5).astype(np.float32)
    B = (rng.randn(16, 32) * 0.5).astype(np.float32)
    tA = torch.from_numpy(A).to(torch.bfloat16).cuda()
    tB = torch.from_numpy(B).to(torch.bfloat16).cuda()
    tD = torch.zeros(32, 32, dtype=torch.float32, device='cuda')

    with open(f'{here}/launch.hip', 'w') as f:
        f.write(SRC + r'''
extern "C" int run(const void* a, const void* b, void* d) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0,
                       (const bf16*)a, (const bf16*)b, (float*)d);
    return (int)hipDeviceSynchronize();
}
extern "C" int run16(const void* a, const void* b, void* d) {
    hipLaunchKernelGGL(probe16, dim3(1), dim3(64), 0, 0,
                       (const bf16*)a, (const bf16*)b, (float*)d);
    return (int)hipDeviceSynchronize();
}
''')
    subprocess.run(['hipcc', '--offload-arch=gfx950', '-O2', '-fPIC',
                    '-shared', f'{here}/launch.hip', '-o',
                    f'{here}/launch.so'], check=True)
    lib = ctypes.CDLL(f'{here}/launch.so')
    rc = lib.run(ctypes.c_void_p(tA.data_ptr()),
                 ctypes.c_void_p(tB.data_ptr()),
                 ctypes.c_void_p(tD.data_ptr()))
    print('rc', rc)
    got = tD.cpu().numpy()
    ref = (tA.float().cpu().numpy() @ tB.float().cpu().numpy())
    err = np.abs(got - ref).max()
    print('max err vs bf16-rounded ref:', err)
    print('match:', err < 0.05)
    if err >= 0.05:
        # localize: check a few entries
        print('got[0,:4]', got[0, :4], 'ref[0,:4]', ref[0, :4])
        print('got[:4,0]', got[:4, 0], 'ref[:4,0]', ref[:4, 0])

    # 16x16x32 variant
    A2 = (rng.randn(16, 32) * 0.5).astype(np.float32)
    B2 = (rng.randn(32, 16) * 0.5).astype(np.float32)
    tA2 = torch.from_numpy(A2).to(torch.bfloat16).cuda()
    tB2 = torch.from_numpy(B2).to(torch.bfloat16).cuda()
    tD2 = torch.zeros(16, 16, dtype=torch.float32, device='cuda')
    rc = lib.run16(ctypes.c_void_p(tA2.data_ptr()),
                   ctypes.c_void_p(tB2.data_ptr()),
                   ctypes.c_void_p(tD2.data_ptr()))
    got2 = tD2.cpu().numpy()
    ref2 = (tA2.float().cpu().numpy() @ tB2.float().cpu().numpy())
    err2 = np.abs(got2 - ref2).max()
    print('16x16x32 rc', rc, 'max err:', err2, 'match:', err2 < 0.05)
    if err2 >= 0.05:
        print('got2[0,:4]', got2[0, :4], 'ref2[0,:4]', ref2[0, :4])
        print('got2[:4,0]', got2[:4, 0], 'ref2[:4,0]', ref2[:4, 0])


if __name__ == '__main__':
    main()
