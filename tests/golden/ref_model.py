"""Seeded 2-layer 3D conv net, reference `pytorch` framework model-file
contract (chunkflow/flow/divid_conquer/patch/pytorch.py:48-60): exposes
`InstantiatedModel`; weights are loaded from the --convnet-weight-path file.
Used as a golden conv-parity pin (torch-CPU reference vs MI355X path)."""
import torch
import torch.nn as nn


class GoldenNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv3d(1, 8, 3, padding=1)
        self.conv2 = nn.Conv3d(8, 3, 3, padding=1)

    def forward(self, x):
        return torch.sigmoid(self.conv2(torch.relu(self.conv1(x))))


InstantiatedModel = GoldenNet()
