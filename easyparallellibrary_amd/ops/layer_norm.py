"""Fused LayerNorm with optional fused residual add (CDNA4 kernel,
fp32 statistics).

Replaces torch's LayerNorm on the hot path (BERT/GPT blocks run it 2x per
layer).  The HIP kernel (csrc/kernels/kernels.hip: layer_norm_fwd/bwd)
streams bf16 rows as packed 16-byte loads with fp32 accumulation, stages
the fp32 row image in LDS between the statistics and normalize passes
(no HBM re-read), and — with a residual input — folds the residual add
into the same pass, eliminating the separate elementwise-add kernel
entirely (post-LN BERT: x = LN(x + sublayer(x))).  The backward caches
xhat / dy*gamma in LDS and flushes dgamma/dbeta once per block.
CPU fallback = identical math in torch (numerics tests compare against a
plain fp32 torch reference).
"""

import torch
import torch.nn as nn

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


def _native_ok(x):
    return (use_native(x) and x.dtype in (torch.bfloat16, torch.float32)
            and x.shape[-1] % 8 == 0)


class _FusedLayerNorm(torch.autograd.Function):
    """y = LN(x (+ residual)).  When residual is given, also returns the
    sum s = x + residual (the pre-LN residual stream)."""

    @staticmethod
    def forward(ctx, x, residual, gamma, beta, eps):
        x = x.contiguous()
        cols = x.shape[-1]
        rows = x.numel() // cols
        has_res = residual is not None
        if has_res:
            # the kernel derives its dtype from x; autocast can hand us
            # a bf16 x with an fp32 residual stream (or vice versa)
            residual = residual.contiguous().to(x.dtype)
        ctx.param_dtype = gamma.dtype
        gamma = gamma.contiguous().to(x.dtype)
        beta = beta.contiguous().to(x.dtype)
        if _native_ok(x):
            mean = torch.empty(rows, dtype=torch.float32, device=x.device)
            rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
            out = torch.empty_like(x)
            s = torch.empty_like(x) if has_res else None
            native_ext().layer_norm_fwd(out, x, residual, s, gamma,
                                        beta, mean, rstd, eps)
            norm_in = s if has_res else x
        else:
            norm_in = (x + residual) if has_res else x
            xf = norm_in.float().reshape(rows, cols)
            mu = xf.mean(dim=1)
            var = xf.var(dim=1, unbiased=False)
            rstd = (var + eps).rsqrt()
            mean = mu
            xhat = (xf - mu[:, None]) * rstd[:, None]
            out = (xhat * gamma.float() + beta.float()).to(x.dtype)
            out = out.reshape(x.shape)
            s = norm_in if has_res else None
        ctx.save_for_backward(norm_in, gamma, mean, rstd)
        ctx.has_res = has_res
        if has_res:
            return out, s
        return out

    @staticmethod
    def backward(ctx, dy, *rest):
        norm_in, gamma, mean, rstd = ctx.saved_tensors
        # with residual, grad also flows through the returned sum s
        ds = rest[0] if (ctx.has_res and rest) else None
        dy = dy.contiguous()
        cols = norm_in.shape[-1]
        rows = norm_in.numel() // cols
        if _native_ok(norm_in):
            dx = torch.empty_like(norm_in)
            dgamma = torch.zeros(cols, dtype=torch.float32,
                                 device=norm_in.device)
            dbeta = torch.zeros(cols, dtype=torch.float32,
                                device=norm_in.device)
            native_ext().layer_norm_bwd(dx, dgamma, dbeta, dy, norm_in,
                                        gamma.contiguous(), mean, rstd)
        else:
            xf = norm_in.float().reshape(rows, cols)
            dyf = dy.float().reshape(rows, cols)
            xhat = (xf - mean[:, None]) * rstd[:, None]
            dyg = dyf * gamma.float()
            c1 = dyg.mean(dim=1, keepdim=True)
            c2 = (dyg * xhat).mean(dim=1, keepdim=True)
            dx = ((dyg - c1 - xhat * c2) * rstd[:, None]).to(norm_in.dtype)
            dx = dx.reshape(norm_in.shape)
            dgamma = (dyf * xhat).sum(dim=0)
            dbeta = dyf.sum(dim=0)
        if ds is not None:
            dx = dx + ds
        if ctx.has_res:
            # d(x) == d(residual) — the add distributes the gradient
            return (dx, dx, dgamma.to(ctx.param_dtype),
                    dbeta.to(ctx.param_dtype), None)
        return (dx, None, dgamma.to(ctx.param_dtype),
                dbeta.to(ctx.param_dtype), None)


class FusedLayerNorm(nn.Module):
    def __init__(self, hidden, eps=1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, x, residual=None):
        """ln(x) — or, with residual, ln(x + residual) (fused).  The
        fused form returns only the normalized output; use
        forward_with_sum when the summed stream is needed (pre-LN)."""
        if residual is None:
            return _FusedLayerNorm.apply(x, None, self.weight, self.bias,
                                         self.eps)
        out, _ = _FusedLayerNorm.apply(x, residual, self.weight, self.bias,
                                       self.eps)
        return out

    def forward_with_sum(self, x, residual):
        """(ln(x+residual), x+residual) — pre-LN blocks keep the sum as
        the residual stream."""
        return _FusedLayerNorm.apply(x, residual, self.weight, self.bias,
                                     self.eps)
