"""Fused flash attention (hand-written CDNA4 kernels, head_dim 64, bf16).

Replaces torch SDPA (AOTriton) on the transformer hot path.  The HIP
kernels (csrc/kernels/attention.hip) keep the whole online softmax in
registers — swapped QK^T so each lane owns one query row, O accumulated
transposed so the rescale is lane-local, P fragments rebuilt with
v_permlane32_swap.  CPU / unsupported shapes fall back to torch SDPA.
"""

import torch
import torch.nn.functional as F

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        out = torch.empty_like(q)
        seq = q.shape[-2]
        bh = q.numel() // (seq * q.shape[-1])
        lse = torch.empty(bh * seq, dtype=torch.float32, device=q.device)
        native_ext().attn_fwd(q, k, v, out, lse, scale, causal)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        dout = dout.contiguous()
        dq = torch.empty_like(q)
        dk = torch.empty_like(k)
        dv = torch.empty_like(v)
        delta = torch.empty_like(lse)
        native_ext().attn_bwd(q, k, v, out, dout, lse, delta, dq, dk, dv,
                              ctx.scale, ctx.causal)
        return dq, dk, dv, None, None


def flash_attention(q, k, v, causal=False, scale=None, allow_native=True):
    """q,k,v: [B, H, S, D].  Native kernel when bf16/D=64 on GPU; torch
    SDPA otherwise."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if (allow_native and use_native(q) and q.dtype == torch.bfloat16
            and q.shape[-1] == 64):
        return _FlashAttention.apply(q, k, v, causal, scale)
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal,
                                          scale=scale)
