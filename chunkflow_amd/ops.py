"""Device-op layer under the Inferencer.

Two implementations of the same small interface over torch tensors:

  * HipOps — the PRODUCT path: hand-written gfx950 kernels through the C-ABI
    extension, stream-ordered on torch's current stream. Raises loudly if
    the extension is missing; never falls back.
  * TorchOps — CPU plumbing for GPU-less machines only (BASELINE config 1 is
    explicitly a no-GPU plumbing config, and the reference itself runs on
    CPU). The Inferencer refuses to pick TorchOps when a GPU is visible, so
    the GPU path can never silently degrade.
"""
import numpy as np
import torch


class HipOps:
    is_hip = True

    def __init__(self, device_index: int = 0):
        from .hip import CfxContext
        self.cfx = CfxContext(device_index)
        self.device = f'cuda:{device_index}'
        self.cfx.adopt_torch_stream()

    def sync(self):
        self.cfx.sync()

    def cast_div(self, u8: torch.Tensor, divisor: float) -> torch.Tensor:
        out = torch.empty(u8.shape, dtype=torch.float32, device=u8.device)
        self.cfx.cast_u8_f32_div(u8.data_ptr(), out.data_ptr(), u8.numel(),
                                 divisor)
        return out

    def normalize_intensity(self, u8: torch.Tensor) -> torch.Tensor:
        out = torch.empty(u8.shape, dtype=torch.float32, device=u8.device)
        self.cfx.normalize_intensity(u8.data_ptr(), out.data_ptr(),
                                     u8.numel())
        return out

    def extract(self, chunk_f32: torch.Tensor, starts: np.ndarray,
                patch_size, out_batch: torch.Tensor):
        dims = chunk_f32.shape[-3:]
        channels = 1 if chunk_f32.ndim == 3 else chunk_f32.shape[0]
        self.cfx.extract_patches(chunk_f32.data_ptr(), channels, dims,
                                 starts, patch_size, out_batch.data_ptr())

    def blend(self, out: torch.Tensor, patch_batch: torch.Tensor,
              batch_index: int, offset, mask: torch.Tensor = None):
        channels = out.shape[0]
        pdims = patch_batch.shape[-3:]
        pvox = pdims[0] * pdims[1] * pdims[2]
        patch_ptr = (patch_batch.data_ptr()
                     + batch_index * patch_batch.shape[1] * pvox * 4)
        self.cfx.blend_accumulate(
            out.data_ptr(), channels, out.shape[-3:], patch_ptr, pdims,
            offset, mask.data_ptr() if mask is not None else None)

    def blend_batch(self, out: torch.Tensor, patch_batch: torch.Tensor,
                    items: np.ndarray, mask: torch.Tensor = None):
        """items: (n, 4) = (batch_index, oz, oy, ox), regions disjoint."""
        channels = out.shape[0]
        self.cfx.blend_batch(
            out.data_ptr(), channels, out.shape[-3:],
            patch_batch.data_ptr(), patch_batch.shape[-3:], items,
            mask.data_ptr() if mask is not None else None)

    def build_chunk_mask(self, out_dims, patch_mask: torch.Tensor,
                         offsets: np.ndarray,
                         groups=None) -> torch.Tensor:
        """Blend the patch mask at every offset then reciprocal. With
        `groups` (disjoint index groups over `offsets`), uses one launch
        per group instead of one per offset."""
        if groups is None:
            mask = torch.empty(tuple(out_dims), dtype=torch.float32,
                               device=patch_mask.device)
            self.cfx.build_chunk_mask(mask.data_ptr(), out_dims,
                                      patch_mask.data_ptr(),
                                      patch_mask.shape[-3:], offsets)
            return mask
        mask = torch.zeros(tuple(out_dims), dtype=torch.float32,
                           device=patch_mask.device)
        mask3 = mask.unsqueeze(0)  # (1, D, H, W): one "channel"
        pm = patch_mask.unsqueeze(0)  # batch of one patch
        for idx in groups:
            items = np.concatenate(
                [np.zeros((len(idx), 1), dtype=np.int32),
                 np.asarray(offsets, dtype=np.int32)[idx]], axis=1)
            self.blend_batch(mask3, pm, items)
        self.cfx.reciprocal(mask.data_ptr(), mask.numel())
        return mask

    def multiply_mask(self, out: torch.Tensor, mask: torch.Tensor):
        self.cfx.multiply_mask(out.data_ptr(), mask.data_ptr(),
                               out.shape[0], mask.numel())

    def multiply_mask_max(self, out: torch.Tensor,
                          mask: torch.Tensor) -> float:
        """Fused mask-normalize + the <1.0001 max scan."""
        m = self.cfx.multiply_mask_max(out.data_ptr(), mask.data_ptr(),
                                       out.shape[0], mask.numel())
        if m is None:
            m = self.cfx.max(out.data_ptr(), out.numel())
        return m

    def max(self, t: torch.Tensor) -> float:
        return self.cfx.max(t.data_ptr(), t.numel())

    def crop_margin(self, t: torch.Tensor, margins) -> torch.Tensor:
        channels = 1 if t.ndim == 3 else t.shape[0]
        d, h, w = t.shape[-3:]
        od = d - margins[0] - margins[3]
        oh = h - margins[1] - margins[4]
        ow = w - margins[2] - margins[5]
        shape = (channels, od, oh, ow) if t.ndim == 4 else (od, oh, ow)
        out = torch.empty(shape, dtype=t.dtype, device=t.device)
        self.cfx.crop_margin(t.data_ptr(), out.data_ptr(), channels,
                             (d, h, w), margins)
        return out

    def mask_using_last_channel(self, t: torch.Tensor,
                                threshold: float) -> torch.Tensor:
        channels = t.shape[0]
        out = torch.empty((channels - 1,) + tuple(t.shape[1:]),
                          dtype=t.dtype, device=t.device)
        self.cfx.mask_using_last_channel(t.data_ptr(), out.data_ptr(),
                                         channels, t.shape[-3:], threshold)
        return out


class TorchOps:
    is_hip = False

    def __init__(self, device_index: int = 0):
        self.device = 'cpu'

    def sync(self):
        pass

    def cast_div(self, u8, divisor):
        return u8.to(torch.float32) / divisor

    def normalize_intensity(self, u8):
        return u8.to(torch.float32) / 127.5 - 1.0

    def extract(self, chunk_f32, starts, patch_size, out_batch):
        pz, py, px = patch_size
        for i, (z0, y0, x0) in enumerate(starts):
            src = chunk_f32[..., z0:z0 + pz, y0:y0 + py, x0:x0 + px]
            out_batch[i] = src if src.ndim == 4 else src.unsqueeze(0)

    def blend(self, out, patch_batch, batch_index, offset, mask=None):
        patch = patch_batch[batch_index]
        if mask is not None:
            patch = patch * mask
        pdims = patch.shape[-3:]
        odims = out.shape[-3:]
        dst, src = [], []
        for off, p, h in zip(offset, pdims, odims):
            lo, hi = max(off, 0), min(off + p, h)
            if hi <= lo:
                return
            dst.append(slice(lo, hi))
            src.append(slice(lo - off, hi - off))
        out[..., dst[0], dst[1], dst[2]] += patch[..., src[0], src[1],
                                                  src[2]]

    def blend_batch(self, out, patch_batch, items, mask=None):
        # CPU plumbing: sequential (order already preserved by grouping)
        for b, oz, oy, ox in items:
            self.blend(out, patch_batch, int(b), (int(oz), int(oy), int(ox)),
                       mask=mask)

    def build_chunk_mask(self, out_dims, patch_mask, offsets, groups=None):
        mask = torch.zeros(tuple(out_dims), dtype=torch.float32)
        pm = patch_mask.unsqueeze(0)  # (1,pz,py,px): one 3-D "patch"
        for off in offsets:
            self.blend(mask, pm, 0, tuple(int(v) for v in off))
        return 1.0 / mask

    def multiply_mask(self, out, mask):
        out *= mask

    def multiply_mask_max(self, out, mask):
        out *= mask
        return out.max().item()

    def max(self, t):
        return t.max().item()

    def crop_margin(self, t, margins):
        d, h, w = t.shape[-3:]
        return t[..., margins[0]:d - margins[3], margins[1]:h - margins[4],
                 margins[2]:w - margins[5]].contiguous()

    def mask_using_last_channel(self, t, threshold):
        keep = t[-1] < threshold
        return t[:-1] * keep
