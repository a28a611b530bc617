"""Identity engine: the test/parity engine (reference patch/identity.py:29-51).

Returns the input patch (f32, channel-cropped, margin-cropped,
channel-repeated to num_output_channels). The bump patch-mask multiply is
fused into the blend kernel (pre_masked=False); the reference multiplies
before the repeat, which is numerically identical per channel.
"""
import torch

from .base import EngineBase


class IdentityEngine(EngineBase):
    pre_masked = False

    def __init__(self, convnet_model, convnet_weight_path, **kw):
        super().__init__(convnet_model, convnet_weight_path, **kw)

    def __call__(self, batch):
        out = batch.to(torch.float32)
        out = self._crop_output_patch(out)
        if self.num_output_channels > 1 and out.shape[1] == 1:
            out = out.expand(-1, self.num_output_channels, -1, -1, -1)
        return out.contiguous()
