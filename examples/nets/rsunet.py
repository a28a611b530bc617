"""3-channel affinity residual 3D U-Net — the benchmark convnet model file.

This file follows the reference's `pytorch` framework model-file contract
(chunkflow/flow/divid_conquer/patch/pytorch.py:48-60): it exposes
`InstantiatedModel`; weights are random-initialized with a fixed seed when no
--convnet-weight-path is given (there is no network for checkpoints —
BASELINE.md measurement plan).

Architecture: a representative residual U-Net for anisotropic EM affinity
prediction (the reference's production workload): widths (28, 36, 48, 64),
(1, 2, 2) pooling so the 20-section z depth is preserved, 3x3x3 residual
blocks with ELU, (1, 5, 5) input/output convs, sigmoid affinity head.
"""
import torch
import torch.nn as nn


class ResBlock(nn.Module):
    def __init__(self, c):
        super().__init__()
        self.conv1 = nn.Conv3d(c, c, 3, padding=1, bias=True)
        self.conv2 = nn.Conv3d(c, c, 3, padding=1, bias=True)
        self.act = nn.ELU(inplace=True)

    def forward(self, x):
        y = self.act(self.conv1(x))
        y = self.conv2(y)
        return self.act(x + y)


class RSUNet(nn.Module):
    WIDTHS = (28, 36, 48, 64)

    def __init__(self, in_channels=1, out_channels=3):
        super().__init__()
        w = self.WIDTHS
        self.conv_in = nn.Conv3d(in_channels, w[0], (1, 5, 5),
                                 padding=(0, 2, 2))
        self.enc = nn.ModuleList([ResBlock(c) for c in w])
        self.down = nn.ModuleList([
            nn.Conv3d(w[i], w[i + 1], (1, 2, 2), stride=(1, 2, 2))
            for i in range(len(w) - 1)])
        self.up = nn.ModuleList([
            nn.ConvTranspose3d(w[i + 1], w[i], (1, 2, 2), stride=(1, 2, 2))
            for i in range(len(w) - 1)])
        self.dec = nn.ModuleList([ResBlock(c) for c in w[:-1]])
        self.conv_out = nn.Conv3d(w[0], out_channels, (1, 5, 5),
                                  padding=(0, 2, 2))

    def forward(self, x):
        x = self.conv_in(x)
        skips = []
        for i, block in enumerate(self.enc):
            x = block(x)
            if i < len(self.down):
                skips.append(x)
                x = self.down[i](x)
        for i in reversed(range(len(self.dec))):
            x = self.up[i](x)
            x = self.dec[i](x + skips[i])
        return torch.sigmoid(self.conv_out(x))


torch.manual_seed(0)
InstantiatedModel = RSUNet()
