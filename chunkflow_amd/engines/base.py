"""Engine base: geometry + patch-mask ownership.

Mirrors the reference PatchInferencerBase contract
(chunkflow/flow/divid_conquer/patch/base.py:11-74): crop margins from the
input/output patch size difference, input overlap/stride derived from the
output ones, and ownership of the bump patch mask. Engines here consume and
produce torch tensors (device-resident for the GPU path); `pre_masked` says
whether the engine already multiplied the patch mask (universal plugins do —
universal.py:22-26) or whether the blend kernel should fuse it.
"""
import torch


class EngineBase:
    pre_masked = False  # blend fuses the patch-mask multiply

    def __init__(self, convnet_model, convnet_weight_path, input_patch_size,
                 output_patch_size, output_patch_overlap, num_input_channels,
                 num_output_channels, patch_mask_np, device, dtype='float32'):
        if output_patch_size is None:
            output_patch_size = input_patch_size
        self.input_patch_size = tuple(input_patch_size)
        self.output_patch_size = tuple(output_patch_size)
        self.output_patch_overlap = tuple(output_patch_overlap)
        self.num_input_channels = num_input_channels
        self.num_output_channels = num_output_channels
        self.dtype = dtype
        self.device = device
        # patch/base.py:27-38
        self.crop_margin = tuple(
            (i - o) // 2 for i, o in zip(input_patch_size, output_patch_size))
        self.input_patch_overlap = tuple(
            o + 2 * c for o, c in zip(output_patch_overlap, self.crop_margin))
        self.input_patch_stride = tuple(
            p - o for p, o in zip(input_patch_size, self.input_patch_overlap))
        self.output_patch_stride = tuple(
            p - o for p, o in
            zip(output_patch_size, self.output_patch_overlap))
        self.patch_mask_np = patch_mask_np  # f32 (pz,py,px), host
        self.patch_mask = torch.from_numpy(patch_mask_np.copy()).to(device)

    @property
    def compute_device(self) -> str:
        if str(self.device).startswith('cuda'):
            return torch.cuda.get_device_name(0)
        import platform
        return platform.processor() or 'cpu'

    def _crop_output_patch(self, patch):
        """channel + symmetric margin crop (patch/base.py:70-74)."""
        cz, cy, cx = self.crop_margin
        return patch[:, :self.num_output_channels,
                     cz:patch.shape[-3] - cz,
                     cy:patch.shape[-2] - cy,
                     cx:patch.shape[-1] - cx]

    def __call__(self, batch):  # (B, C_in, pz, py, px) torch tensor
        raise NotImplementedError
