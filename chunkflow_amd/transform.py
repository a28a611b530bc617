"""Test-time-augmentation transforms (the `--augment` flag).

Replicates the OUTPUT behavior of the reference's
chunkflow/flow/divid_conquer/transform.py:114-156: 8 sequences from
(Lazy|Transpose) x (Lazy|FlipLR) x (Lazy|FlipUD), forward-applied before the
engine and backward-applied (same order) after, then averaged.

Reference quirk (replicated, documented in DESIGN.md): the reference applies
np.fliplr/np.flipud to the 4-D (B, C, y, x) slice at each z, which flips the
CHANNEL (FlipLR) and BATCH (FlipUD) axes — not y/x. Only Transpose is a real
spatial transform. The round trip is exact for batch-independent engines, so
the averaged output matches the reference bit-for-bit in distribution; we do
the same dims here.
"""
from itertools import product

import torch


class _Lazy:
    def forward(self, x):
        return x

    backward = forward


class _Transpose:
    def forward(self, x):
        return x.swapaxes(-1, -2)

    backward = forward


class _FlipChannel:  # the reference's 'FlipLR' on a 5-D patch
    def forward(self, x):
        return torch.flip(x, dims=[1]) if isinstance(x, torch.Tensor) \
            else x[:, ::-1].copy()

    backward = forward


class _FlipBatch:  # the reference's 'FlipUD' on a 5-D patch
    def forward(self, x):
        return torch.flip(x, dims=[0]) if isinstance(x, torch.Tensor) \
            else x[::-1].copy()

    backward = forward


class TransformSequences:
    def __init__(self, transpose=True, fliplr=True, flipud=True):
        options = []
        if transpose:
            options.append((_Lazy(), _Transpose()))
        if fliplr:
            options.append((_Lazy(), _FlipChannel()))
        if flipud:
            options.append((_Lazy(), _FlipBatch()))
        assert options
        self.sequences = list(product(*options))

    def forward(self, patch):
        out = []
        for seq in self.sequences:
            t = patch
            for tr in seq:
                t = tr.forward(t)
            out.append(t)
        return out

    def backward(self, patches):
        assert len(patches) == len(self.sequences)
        out = []
        for seq, p in zip(self.sequences, patches):
            for tr in seq:
                p = tr.backward(p)
            out.append(p)
        return out
