"""PyTorch engine: run a user model file's net on the device batch.

Keeps the reference model-file contract
(chunkflow/flow/divid_conquer/patch/pytorch.py:48-83): the file exposes
`InstantiatedModel` (weights loaded from --convnet-weight-path via
torch.load, 'state_dict' key unwrapped) or `load_model(weight_path)`, plus
optional `pre_process` / `post_process` hooks.

MI355X-first differences from the reference engine:
  * the batch STAYS on the GPU — no per-patch H2D/D2H round trip
    (the reference moves every patch to cpu numpy, pytorch.py:115-118);
  * batch_size > 1 is allowed (the reference hard-asserts 1,
    inferencer.py:217-219; parity runs compare at batch 1);
  * user pre_process/post_process hooks receive torch tensors (on device),
    not numpy arrays.
The patch-mask multiply (pytorch.py:113) is fused into the blend kernel.
"""
import os

import torch

from .base import EngineBase
from ..model_loader import load_source


class PyTorchEngine(EngineBase):
    pre_masked = False

    def __init__(self, convnet_model, convnet_weight_path, **kw):
        super().__init__(convnet_model, convnet_weight_path, **kw)
        net_source = load_source(convnet_model)
        if hasattr(net_source, 'load_model'):
            self.model = net_source.load_model(convnet_weight_path)
        else:
            self.model = net_source.InstantiatedModel
            if convnet_weight_path is not None:
                chkpt = torch.load(convnet_weight_path,
                                   map_location='cpu')
                state_dict = chkpt.get('state_dict', chkpt) \
                    if isinstance(chkpt, dict) else chkpt
                self.model.load_state_dict(state_dict)
        self.model = self.model.to(self.device)
        self.model.eval()
        if self.dtype == 'bfloat16':
            self.model = self.model.to(torch.bfloat16)
        # NDHWC keeps MIOpen on its fast implicit-GEMM solvers (measured
        # 78.6 vs 112 ms per 12-patch RSUNet batch on MI355X); disable with
        # CFX_CHANNELS_LAST=0. benchmark=True matches the reference
        # (pytorch.py:7) and lets MIOpen tune per shape.
        self.channels_last = (str(self.device).startswith('cuda') and
                              os.environ.get('CFX_CHANNELS_LAST', '1') != '0')
        if self.channels_last:
            self.model = self.model.to(memory_format=torch.channels_last_3d)
            # swap eligible 3x3x3 ResBlock convs for the hand-written MFMA
            # kernels (fastconv.py); CFX_FASTCONV=0 keeps MIOpen everywhere
            if os.environ.get('CFX_FASTCONV', '1') != '0':
                idx = int(str(self.device).split(':')[-1])                     if ':' in str(self.device) else 0
                from ..fastconv import (accelerate_conv_in,
                                        accelerate_updown)
                bf = self.dtype == 'bfloat16'
                if bf:
                    from ..fastconv import maybe_accelerate_bf16
                    self.fastconv_count = maybe_accelerate_bf16(
                        self.model, idx)
                else:
                    from ..fastconv import maybe_accelerate
                    self.fastconv_count = maybe_accelerate(self.model, idx)
                self.fastconv_count += accelerate_updown(
                    self.model, idx, bf16=bf)
                self.fastconv_count += accelerate_conv_in(
                    self.model, idx, bf16=bf)
                # conv_out kernel beats MIOpen at both dtypes (4.8x bf16,
                # 1.4x f32 -- profiles/updown_probe_r02.json)
                from ..fastconv import accelerate_conv_out
                self.fastconv_count += accelerate_conv_out(
                    self.model, idx, bf16=bf)
        torch.backends.cudnn.benchmark = True
        self.pre_process = getattr(net_source, 'pre_process', None)
        self.post_process = getattr(net_source, 'post_process', None)

    @torch.no_grad()
    def __call__(self, batch):
        x = batch
        if self.pre_process is not None:
            x = self.pre_process(x)
        if self.dtype == 'bfloat16':
            x = x.to(torch.bfloat16)
        if self.channels_last:
            x = x.contiguous(memory_format=torch.channels_last_3d)
        out = self.model(x)
        if self.post_process is not None:
            out = self.post_process(out)
        if out.dtype != torch.float32:
            out = out.to(torch.float32)
        out = self._crop_output_patch(out)
        return out.contiguous()
