"""Fused Atari stem: u8 dequant + first conv with a hand-written
gfx950 weight-gradient kernel.

The IMPALA/APEX learner's dominant kernel is the weight gradient of
Conv2d(4, 32, 8, stride=4) — a skinny reduction GEMM MIOpen runs at
~2% MFMA efficiency (profiles/). :class:`FusedAtariConv1` computes:

* forward: u8 frames -> bf16 dequant (fused kernel) -> MIOpen conv;
* backward: grad_weight by ``ops/hip/conv1_wrw.hip`` (MFMA split-K
  over im2col rows assembled straight from the u8 frames — x is read
  ONCE as uint8); grad_bias by a plain reduction; grad_input skipped
  (frames are data).
"""
import torch as t
import torch.nn as nn
import torch.nn.functional as F

from . import _require_ext, dequant_u8


class _FusedConv1Fn(t.autograd.Function):
    @staticmethod
    def forward(ctx, frames_u8: t.Tensor, weight: t.Tensor,
                bias: t.Tensor, scale: float):
        # frames_u8: [B, 4, 84, 84] channels_last uint8
        ext = _require_ext()
        B = frames_u8.shape[0]
        nhwc = frames_u8.permute(0, 2, 3, 1).contiguous()
        # patch-major weight repack: n = r*32 + c*4 + ci
        w_rs = (
            weight.permute(2, 3, 1, 0).reshape(256, 32)
            .to(t.bfloat16).contiguous()
        )
        y_rows = ext.conv1_fwd(
            nhwc, w_rs,
            bias.float().contiguous() if bias is not None
            else t.empty(0),
            scale,
        )
        # [K,32] IS the NHWC image; permute -> channels_last NCHW
        y = y_rows.view(B, 20, 20, 32).permute(0, 3, 1, 2)
        ctx.save_for_backward(frames_u8, weight)
        ctx.scale = scale
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy: t.Tensor):
        frames_u8, weight = ctx.saved_tensors
        ext = _require_ext()
        gy_cl = gy.contiguous(memory_format=t.channels_last)
        dy_rows = gy_cl.permute(0, 2, 3, 1).reshape(-1, 32)
        # channels_last u8 frames: permute exposes the NHWC memory as
        # a standard-contiguous view (no copy)
        frames_nhwc = frames_u8.permute(0, 2, 3, 1).contiguous()
        grad_w, grad_b = ext.conv1_wrw(
            dy_rows.to(t.bfloat16).contiguous(), frames_nhwc, ctx.scale
        )
        grad_w = grad_w.to(weight.dtype)
        if not ctx.has_bias:
            grad_b = None
        return None, grad_w, grad_b, None


class FusedAtariConv1(nn.Module):
    """Drop-in for the Atari stem conv, consuming RAW uint8 frames
    (channels_last) and emitting bf16 activations."""

    def __init__(self, scale: float = 1.0 / 255.0):
        super().__init__()
        ref = nn.Conv2d(4, 32, 8, stride=4)
        self.weight = nn.Parameter(ref.weight.detach().clone())
        self.bias = nn.Parameter(ref.bias.detach().clone())
        self.scale = scale

    def forward(self, frames_u8: t.Tensor) -> t.Tensor:
        return _FusedConv1Fn.apply(
            frames_u8, self.weight, self.bias, self.scale
        )
