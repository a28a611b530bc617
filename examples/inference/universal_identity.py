"""Example `--framework universal` model file (the reference contract,
examples/inference/universal_identity.py there): PatchInferencer takes the
weight file and the bump patch mask, and returns ALREADY-masked float32
patches."""
import platform

import numpy as np


class PatchInferencer:
    def __init__(self, model_weight_file, output_patch_mask):
        self.output_patch_mask = output_patch_mask

    @property
    def compute_device(self):
        return platform.processor() or 'cpu'

    def __call__(self, input_patch):
        # input: (B, C, z, y, x) float32 in [0, 1]
        output_patch = np.repeat(input_patch.astype(np.float32), 3, axis=1)
        return output_patch * self.output_patch_mask
