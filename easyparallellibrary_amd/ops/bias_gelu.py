"""Fused bias + GeLU (tanh approximation), CDNA4 kernel.

No reference counterpart (plain TF ops there); MI355X-native fusion for
the HBM-bound FFN epilogue.

The FFN's first Linear runs without bias; the bias-add and activation fuse
into one streaming kernel pass (saves two full activation read/writes per
FFN on the 8 TB/s HBM bound).  CPU fallback in torch.
"""

import torch
import torch.nn as nn

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


def _gelu_tanh(x):
    return 0.5 * x * (1.0 + torch.tanh(
        0.7978845608028654 * (x + 0.044715 * x * x * x)))


class _FusedBiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        x = x.contiguous()
        cols = x.shape[-1]
        # under autocast the activation is bf16 while the bias PARAM
        # stays fp32 — the kernel derives its dtype from x, so cast the
        # bias (param-dtype grads are restored in backward)
        ctx.param_dtype = bias.dtype
        kbias = bias.contiguous().to(x.dtype)
        if use_native(x) and x.dtype in (torch.bfloat16, torch.float32) \
                and cols % 8 == 0:
            out = torch.empty_like(x)
            native_ext().bias_gelu_fwd(out, x, kbias)
        else:
            out = _gelu_tanh((x.float() + kbias.float())).to(x.dtype)
        ctx.save_for_backward(x, kbias)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        dy = dy.contiguous()
        cols = x.shape[-1]
        if use_native(x) and x.dtype in (torch.bfloat16, torch.float32) \
                and cols % 8 == 0:
            dx = torch.empty_like(x)
            dbias = torch.zeros(cols, dtype=torch.float32, device=x.device)
            native_ext().bias_gelu_bwd(dx, dbias, dy, x, bias)
        else:
            xb = x.float() + bias.float()
            t = torch.tanh(0.7978845608028654 * (xb + 0.044715 * xb ** 3))
            dgelu = 0.5 * (1 + t) + 0.5 * xb * (1 - t * t) * \
                0.7978845608028654 * (1 + 3 * 0.044715 * xb * xb)
            dx = (dy.float() * dgelu).to(x.dtype)
            dbias = dx.float().reshape(-1, cols).sum(dim=0)
        return dx, dbias.to(ctx.param_dtype)


class FusedBiasGelu(nn.Module):
    def __init__(self, hidden):
        super().__init__()
        self.bias = nn.Parameter(torch.zeros(hidden))

    def forward(self, x):
        return _FusedBiasGelu.apply(x, self.bias)
