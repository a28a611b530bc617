"""Graph surgery: swap eligible nn.Conv3d modules for the hand-written
gfx950 MFMA conv kernel (csrc/conv.hip).

Eligible: kernel (3,3,3), stride 1, padding 1, dilation 1, groups 1,
in_channels == out_channels in {28, 36, 48, 64} — the RSUNet ResBlock
convs, ~85% of the affinity UNet's FLOPs. Everything else stays on
MIOpen. Enabled in the pytorch engine via CFX_FASTCONV (see engine);
weights are repacked once to the kernel's (27, C, K) tap-major layout.
"""
import os

import torch
import torch.nn as nn

_CTX = {}


def get_cfx(device_index: int = 0):
    if device_index not in _CTX:
        from .hip import CfxContext
        ctx = CfxContext(device_index)
        ctx.adopt_torch_stream()
        _CTX[device_index] = ctx
    return _CTX[device_index]


# widths where the hand kernel BEATS MIOpen (the persistent-z ring:
# C=28 88.6 vs 81.9 TF, C=36 69.5 vs 64.7; 48/64 stay on MIOpen — the
# weight wall no longer fits LDS beside the ring at those widths, and
# their spatial extents are small — DESIGN.md §10 ladder)
ELIGIBLE_WIDTHS = (28, 36)
ZRING_WIDTHS = (28, 36)


def _eligible(m: nn.Module) -> bool:
    return (isinstance(m, nn.Conv3d)
            and m.kernel_size == (3, 3, 3)
            and m.stride == (1, 1, 1)
            and m.padding == (1, 1, 1)
            and m.dilation == (1, 1, 1)
            and m.groups == 1
            and m.in_channels == m.out_channels
            and m.in_channels in ELIGIBLE_WIDTHS)


class CfxConv3d(nn.Module):
    """Drop-in for an eligible nn.Conv3d; consumes/produces channels-last
    (NDHWC) tensors, exactly the memory format the engine runs in."""

    def __init__(self, conv: nn.Conv3d, device_index: int = 0):
        super().__init__()
        self.C = conv.in_channels
        self.K = conv.out_channels
        self.device_index = device_index
        w = conv.weight.detach()  # (K, C, 3, 3, 3)
        wtap = w.permute(2, 3, 4, 1, 0).reshape(27, self.C, self.K)
        self.register_buffer('wtap', wtap.contiguous().float())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def _run(self, x, residual=None, elu=False):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        out = torch.empty((n, self.K, d, h, w), dtype=torch.float32,
                          device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).conv3_ndhwc(
            x.data_ptr(), self.wtap.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            residual.data_ptr() if residual is not None else None,
            out.data_ptr(), n, d, h, w, self.C, self.K, do_elu=elu,
            zring=self.C in ZRING_WIDTHS)
        return out

    def forward(self, x):
        assert x.dtype == torch.float32, 'fastconv is the f32 path'
        return self._run(x)


class CfxResBlock(nn.Module):
    """Fused replacement for the RSUNet ResBlock pattern (conv1 -> ELU ->
    conv2 -> +x -> ELU): the ELU and residual-add run in the conv
    epilogues, removing three full-tensor elementwise passes per block."""

    def __init__(self, block: nn.Module, device_index: int = 0):
        super().__init__()
        self.c1 = CfxConv3d(block.conv1, device_index)
        self.c2 = CfxConv3d(block.conv2, device_index)

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        h = self.c1._run(x, elu=True)
        return self.c2._run(h, residual=x, elu=True)


def _resblock_like(m: nn.Module) -> bool:
    return (hasattr(m, 'conv1') and hasattr(m, 'conv2')
            and isinstance(getattr(m, 'act', None), nn.ELU)
            and getattr(m.act, 'alpha', None) == 1.0
            and not isinstance(m, (CfxConv3d, CfxResBlock))
            and _eligible(m.conv1) and _eligible(m.conv2))


@torch.no_grad()
def _fusion_matches(block: nn.Module, fused: 'CfxResBlock') -> bool:
    """Functional probe at surgery time: the fused block must reproduce the
    original forward on a random input (guards against user modules that
    merely LOOK like a ResBlock)."""
    C = fused.c1.C
    x = torch.randn(1, C, 4, 18, 22, device=fused.c1.wtap.device)         .contiguous(memory_format=torch.channels_last_3d)
    want = block.to(x.device)(x)
    got = fused(x)
    return bool(torch.allclose(got, want, rtol=1e-4, atol=1e-4))


def maybe_accelerate(model: nn.Module, device_index: int = 0) -> int:
    """Replace eligible ResBlocks (fused) and lone convs in-place;
    returns the replacement count."""
    count = 0
    dev = f'cuda:{device_index}'
    fused_out = set()  # detached originals (their convs must not re-count)
    for parent in list(model.modules()):  # snapshot: we mutate the tree
        if id(parent) in fused_out:
            continue
        for name, child in list(parent.named_children()):
            if _resblock_like(child):
                fused = CfxResBlock(child, device_index).to(dev)
                if _fusion_matches(child, fused):
                    setattr(parent, name, fused)
                    fused_out.add(id(child))
                    count += 1
                    continue
                # forward does something else: fall through to lone convs
                for cn in ('conv1', 'conv2'):
                    setattr(child, cn,
                            CfxConv3d(getattr(child, cn),
                                      device_index).to(dev))
                    count += 1
            elif _eligible(child):
                setattr(parent, name,
                        CfxConv3d(child, device_index).to(dev))
                count += 1
    return count


# bf16 ring widths on the hand kernels (measured,
# profiles/updown_probe_r02.json): C=28 594-624 TF (4x MIOpen), C=36
# sliced 249 TF (1.31x). C=48 sliced reaches 318 vs MIOpen's 344 raw and
# even with the fused ResBlock epilogue the same-box bench A/B reads
# 292.4/293.5 (C48 on) vs 297.1/296.5 (off) MV/s -- measured-rejected,
# kernel kept callable behind CFX_BF16_WIDTHS=28,36,48.
BF16_WIDTHS = tuple(
    int(w) for w in os.environ.get('CFX_BF16_WIDTHS', '28,36').split(',')
    if w)


class CfxConv3dBF16(nn.Module):
    """bf16 drop-in for an eligible nn.Conv3d (config-5 path): the bf16
    persistent-z ring kernel on v_mfma_f32_32x32x16_bf16."""

    def __init__(self, conv: nn.Conv3d, device_index: int = 0):
        super().__init__()
        self.C = conv.in_channels
        self.K = conv.out_channels
        self.device_index = device_index
        w = conv.weight.detach().float()  # (K, C, 3, 3, 3)
        # C <= 32 uses the (27, 32, 32) pack of the single-tile ring;
        # 36/48 use the (27, 64, 48) pack of the sliced 4-launch schedule
        if self.C <= 32:
            pack = torch.zeros(27, 32, 32)
        else:
            pack = torch.zeros(27, 64, 48)
        pack[:, :self.K, :self.C] = w.permute(2, 3, 4, 0, 1) \
            .reshape(27, self.K, self.C)
        self.register_buffer('wpack', pack.to(torch.bfloat16).contiguous())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def _run(self, x, residual=None, elu=False):
        assert x.dtype == torch.bfloat16
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        out = torch.empty((n, self.K, d, h, w), dtype=torch.bfloat16,
                          device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).conv3_ndhwc_bf16(
            x.data_ptr(), self.wpack.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            residual.data_ptr() if residual is not None else None,
            out.data_ptr(), n, d, h, w, self.C, self.K, do_elu=elu)
        return out

    def forward(self, x):
        return self._run(x)


class CfxResBlockBF16(nn.Module):
    def __init__(self, block: nn.Module, device_index: int = 0):
        super().__init__()
        self.c1 = CfxConv3dBF16(block.conv1, device_index)
        self.c2 = CfxConv3dBF16(block.conv2, device_index)

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        h = self.c1._run(x, elu=True)
        return self.c2._run(h, residual=x, elu=True)


def maybe_accelerate_bf16(model: nn.Module, device_index: int = 0) -> int:
    """bf16 surgery (widths where the bf16 ring is instantiated); same
    functional-probe guard as the f32 path."""
    count = 0
    dev = f'cuda:{device_index}'

    def elig(m):
        return (isinstance(m, nn.Conv3d) and m.kernel_size == (3, 3, 3)
                and m.stride == (1, 1, 1) and m.padding == (1, 1, 1)
                and m.dilation == (1, 1, 1) and m.groups == 1
                and m.in_channels == m.out_channels
                and m.in_channels in BF16_WIDTHS)

    def rb(m):
        return (hasattr(m, 'conv1') and hasattr(m, 'conv2')
                and isinstance(getattr(m, 'act', None), nn.ELU)
                and getattr(m.act, 'alpha', None) == 1.0
                and not isinstance(m, (CfxConv3dBF16, CfxResBlockBF16))
                and elig(m.conv1) and elig(m.conv2))

    @torch.no_grad()
    def matches(block, fused):
        x = torch.randn(1, fused.c1.C, 4, 18, 22, device=dev) \
            .to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last_3d)
        want = block.to(dev)(x)
        got = fused(x)
        return bool(torch.allclose(got.float(), want.float(), rtol=0.05,
                                   atol=0.05))

    fused_out = set()
    for parent in list(model.modules()):  # snapshot: we mutate the tree
        if id(parent) in fused_out:
            continue
        for name, child in list(parent.named_children()):
            if rb(child):
                fused = CfxResBlockBF16(child, device_index).to(dev)
                if matches(child, fused):
                    setattr(parent, name, fused)
                    fused_out.add(id(child))
                    count += 1
                    continue
                for cn in ('conv1', 'conv2'):
                    setattr(child, cn,
                            CfxConv3dBF16(getattr(child, cn),
                                          device_index).to(dev))
                    count += 1
            elif elig(child):
                setattr(parent, name,
                        CfxConv3dBF16(child, device_index).to(dev))
                count += 1
    return count


# --------------------------------------------------------------------------
# up/down-sampling convs: the RSUNet (1,2,2)-kernel, (1,2,2)-stride shapes
# are HBM-streaming work; csrc/updown.hip replaces MIOpen's implicit GEMM
# (bf16 bwd_data runs ~20x below the stream roofline there).
# --------------------------------------------------------------------------
class CfxUpConv3d(nn.Module):
    """Drop-in for nn.ConvTranspose3d(kernel=(1,2,2), stride=(1,2,2))."""

    def __init__(self, conv: nn.ConvTranspose3d, device_index: int = 0,
                 bf16: bool = False):
        super().__init__()
        self.C = conv.in_channels
        self.K = conv.out_channels
        self.device_index = device_index
        self.bf16 = bf16
        w = conv.weight.detach().float()      # (C, K, 1, 2, 2)
        pack = w[:, :, 0].permute(2, 3, 0, 1).reshape(4, self.C, self.K)
        dt = torch.bfloat16 if bf16 else torch.float32
        self.register_buffer('wpack', pack.to(dt).contiguous())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        dt = torch.bfloat16 if self.bf16 else torch.float32
        assert x.dtype == dt
        out = torch.empty((n, self.K, d, 2 * h, 2 * w), dtype=dt,
                          device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).upconv_2x2(
            x.data_ptr(), self.wpack.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            out.data_ptr(), n, d, h, w, self.C, self.K, bf16=self.bf16)
        return out


class CfxDownConv3d(nn.Module):
    """Drop-in for nn.Conv3d(kernel=(1,2,2), stride=(1,2,2))."""

    def __init__(self, conv: nn.Conv3d, device_index: int = 0,
                 bf16: bool = False):
        super().__init__()
        self.C = conv.in_channels
        self.K = conv.out_channels
        self.device_index = device_index
        self.bf16 = bf16
        w = conv.weight.detach().float()      # (K, C, 1, 2, 2)
        pack = w[:, :, 0].permute(2, 3, 1, 0).reshape(4, self.C, self.K)
        dt = torch.bfloat16 if bf16 else torch.float32
        self.register_buffer('wpack', pack.to(dt).contiguous())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        dt = torch.bfloat16 if self.bf16 else torch.float32
        assert x.dtype == dt
        out = torch.empty((n, self.K, d, h // 2, w // 2), dtype=dt,
                          device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).downconv_2x2(
            x.data_ptr(), self.wpack.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            out.data_ptr(), n, d, h, w, self.C, self.K, bf16=self.bf16)
        return out


# surgery allowlist (measured, profiles/updown_probe_r02.json): the hand
# up-conv wins at in_channels 36 (7.1x bf16 / 2.1x f32) and 48 (tie bf16 /
# 1.4x f32); it loses at 64 (32^2 extent) and the down shapes lose to
# MIOpen everywhere (0.03-0.25x) -- those keep MIOpen; kernels stay
# callable and GPU-tested for the record.
UP_SURGERY_WIDTHS = (36, 48)


def _up_eligible(m) -> bool:
    return (isinstance(m, nn.ConvTranspose3d)
            and m.kernel_size == (1, 2, 2) and m.stride == (1, 2, 2)
            and m.padding == (0, 0, 0) and m.output_padding == (0, 0, 0)
            and m.dilation == (1, 1, 1) and m.groups == 1
            and m.in_channels in UP_SURGERY_WIDTHS
            and m.out_channels <= 64)


def _down_eligible(m) -> bool:
    return (isinstance(m, nn.Conv3d)
            and m.kernel_size == (1, 2, 2) and m.stride == (1, 2, 2)
            and m.padding == (0, 0, 0) and m.dilation == (1, 1, 1)
            and m.groups == 1
            and m.in_channels <= 64 and m.out_channels <= 64)


@torch.no_grad()
def _updown_matches(orig, repl, dev, bf16):
    x = torch.randn(1, repl.C, 3, 12, 16, device=dev)
    if bf16:
        x = x.to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last_3d)
    want = orig.to(dev)(x).float()
    got = repl(x).float()
    tol = 0.05 if bf16 else 1e-4
    return bool(torch.allclose(got, want, rtol=tol, atol=tol))


def accelerate_updown(model: nn.Module, device_index: int = 0,
                      bf16: bool = False) -> int:
    """Swap eligible up/down-sampling convs in-place; returns count."""
    count = 0
    dev = f'cuda:{device_index}'
    for parent in list(model.modules()):
        for name, child in list(parent.named_children()):
            repl = None
            if _up_eligible(child):
                repl = CfxUpConv3d(child, device_index, bf16=bf16).to(dev)
            if repl is not None and _updown_matches(child, repl, dev, bf16):
                setattr(parent, name, repl)
                count += 1
    return count


class CfxConvIn155(nn.Module):
    """Drop-in for nn.Conv3d(1, K, (1,5,5), padding=(0,2,2)) — the RSUNet
    input conv. MIOpen's implicit GEMM degenerates on the single input
    channel (measured 38.8 ms bf16 / 6.8 ms f32 per batch-24 launch vs
    ~0.5 ms algorithmic); csrc/updown.hip runs it as a 2-D stencil."""

    def __init__(self, conv: nn.Conv3d, device_index: int = 0,
                 bf16: bool = False):
        super().__init__()
        self.K = conv.out_channels
        self.device_index = device_index
        self.bf16 = bf16
        w = conv.weight.detach().float()      # (K, 1, 1, 5, 5)
        dt = torch.bfloat16 if bf16 else torch.float32
        self.register_buffer('wpack',
                             w.reshape(self.K, 25).to(dt).contiguous())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        assert c == 1
        dt = torch.bfloat16 if self.bf16 else torch.float32
        assert x.dtype == dt
        out = torch.empty((n, self.K, d, h, w), dtype=dt, device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).conv155_c1(
            x.data_ptr(), self.wpack.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            out.data_ptr(), n, d, h, w, self.K, bf16=self.bf16)
        return out


def _conv155_eligible(m) -> bool:
    return (isinstance(m, nn.Conv3d) and m.kernel_size == (1, 5, 5)
            and m.stride == (1, 1, 1) and m.padding == (0, 2, 2)
            and m.dilation == (1, 1, 1) and m.groups == 1
            and m.in_channels == 1 and m.out_channels <= 32)


@torch.no_grad()
def _conv155_matches(orig, repl, dev, bf16):
    x = torch.randn(1, 1, 3, 14, 19, device=dev)
    if bf16:
        x = x.to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last_3d)
    want = orig.to(dev)(x).float()
    got = repl(x).float()
    tol = 0.05 if bf16 else 1e-4
    return bool(torch.allclose(got, want, rtol=tol, atol=tol))


def accelerate_conv_in(model: nn.Module, device_index: int = 0,
                       bf16: bool = False) -> int:
    """Swap the eligible single-channel (1,5,5) input convs; returns
    count."""
    count = 0
    dev = f'cuda:{device_index}'
    for parent in list(model.modules()):
        for name, child in list(parent.named_children()):
            if _conv155_eligible(child):
                repl = CfxConvIn155(child, device_index, bf16=bf16).to(dev)
                if _conv155_matches(child, repl, dev, bf16):
                    setattr(parent, name, repl)
                    count += 1
    return count


class CfxConvOut155(nn.Module):
    """Drop-in for nn.Conv3d(C, K<=3-ish, (1,5,5), padding=(0,2,2)) — the
    RSUNet output conv. MIOpen's bf16 implicit GEMM degenerates on the
    tiny K (measured 38.6 ms per batch-24 launch vs ~2 ms here)."""

    def __init__(self, conv: nn.Conv3d, device_index: int = 0,
                 bf16: bool = False):
        super().__init__()
        self.C = conv.in_channels
        self.K = conv.out_channels
        self.device_index = device_index
        self.bf16 = bf16
        w = conv.weight.detach().float()      # (K, C, 1, 5, 5)
        pack = w[:, :, 0].permute(0, 2, 3, 1).reshape(self.K, 25, self.C)
        dt = torch.bfloat16 if bf16 else torch.float32
        self.register_buffer('wpack', pack.to(dt).contiguous())
        if conv.bias is not None:
            self.register_buffer('bias', conv.bias.detach().float())
        else:
            self.bias = None

    def forward(self, x):
        x = x.contiguous(memory_format=torch.channels_last_3d)
        n, c, d, h, w = x.shape
        dt = torch.bfloat16 if self.bf16 else torch.float32
        assert x.dtype == dt and c == self.C
        out = torch.empty((n, self.K, d, h, w), dtype=dt, device=x.device,
                          memory_format=torch.channels_last_3d)
        get_cfx(self.device_index).conv155_out(
            x.data_ptr(), self.wpack.data_ptr(),
            self.bias.data_ptr() if self.bias is not None else None,
            out.data_ptr(), n, d, h, w, self.C, self.K, bf16=self.bf16)
        return out


def _conv155_out_eligible(m) -> bool:
    return (isinstance(m, nn.Conv3d) and m.kernel_size == (1, 5, 5)
            and m.stride == (1, 1, 1) and m.padding == (0, 2, 2)
            and m.dilation == (1, 1, 1) and m.groups == 1
            and m.in_channels == 28 and m.out_channels == 3)


def accelerate_conv_out(model: nn.Module, device_index: int = 0,
                        bf16: bool = False) -> int:
    """Swap the eligible few-channel (1,5,5) output convs (bf16 wins big;
    f32 MIOpen is already fine there — callers gate)."""
    count = 0
    dev = f'cuda:{device_index}'
    for parent in list(model.modules()):
        for name, child in list(parent.named_children()):
            if _conv155_out_eligible(child):
                repl = CfxConvOut155(child, device_index, bf16=bf16).to(dev)
                if _conv155_matches_generic(child, repl, dev, bf16,
                                            channels=repl.C):
                    setattr(parent, name, repl)
                    count += 1
    return count


@torch.no_grad()
def _conv155_matches_generic(orig, repl, dev, bf16, channels=1):
    x = torch.randn(1, channels, 3, 14, 19, device=dev)
    if bf16:
        x = x.to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last_3d)
    want = orig.to(dev)(x).float()
    got = repl(x).float()
    tol = 0.05 if bf16 else 1e-4
    return bool(torch.allclose(got, want, rtol=tol, atol=tol))
