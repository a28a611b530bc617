"""Universal engine: delegate the whole patch forward (and masking) to a
user file's PatchInferencer.

Keeps the reference contract EXACTLY
(chunkflow/flow/divid_conquer/patch/universal.py:28-69 and
examples/inference/universal_identity.py in the reference): the user file
defines `class PatchInferencer` with `__init__(model_weight_file,
output_patch_mask)` and `__call__(input_patch: f32 ndarray (B,C,pz,py,px) in
[0,1]) -> f32 ndarray, ALREADY masked`, optional `.compute_device`.

Because the plugin speaks numpy, this engine crosses the device boundary per
batch (D2H in, H2D out) — the price of the unchanged plugin API; the
built-in engines stay device-resident.
"""
import numpy as np
import torch

from .base import EngineBase
from ..model_loader import load_source


class UniversalEngine(EngineBase):
    pre_masked = True  # the plugin output is already masked

    def __init__(self, convnet_model, convnet_weight_path, **kw):
        super().__init__(convnet_model, convnet_weight_path, **kw)
        net_source = load_source(convnet_model)
        assert hasattr(net_source, 'PatchInferencer'), \
            'universal model file must define class PatchInferencer'
        self.patch_inferencer = net_source.PatchInferencer(
            convnet_weight_path, self.patch_mask_np)

    @property
    def compute_device(self):
        if hasattr(self.patch_inferencer, 'compute_device'):
            return self.patch_inferencer.compute_device
        return super().compute_device

    def __call__(self, batch):
        np_in = batch.detach().cpu().numpy()
        np_out = self.patch_inferencer(np_in)
        assert isinstance(np_out, np.ndarray)
        out = torch.from_numpy(np.ascontiguousarray(np_out)).to(self.device)
        return out
