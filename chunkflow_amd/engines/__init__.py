"""Patch inference engines: identity (test oracle), pytorch (model file),
universal (user PatchInferencer plugin). See each module for the reference
contract it keeps."""
from .identity import IdentityEngine
from .pytorch import PyTorchEngine
from .universal import UniversalEngine


def create_engine(framework: str, convnet_model, convnet_weight_path,
                  input_patch_size, output_patch_size, output_patch_overlap,
                  num_input_channels, num_output_channels, patch_mask_np,
                  device, dtype='float32', bump='wu'):
    assert bump == 'wu', 'only the wu bump function is supported'
    kw = dict(input_patch_size=input_patch_size,
              output_patch_size=output_patch_size,
              output_patch_overlap=output_patch_overlap,
              num_input_channels=num_input_channels,
              num_output_channels=num_output_channels,
              patch_mask_np=patch_mask_np, device=device, dtype=dtype)
    if framework == 'identity':
        return IdentityEngine(convnet_model, convnet_weight_path, **kw)
    if framework == 'pytorch':
        return PyTorchEngine(convnet_model, convnet_weight_path, **kw)
    if framework == 'universal':
        return UniversalEngine(convnet_model, convnet_weight_path, **kw)
    raise ValueError(f'invalid inference framework: {framework}')
