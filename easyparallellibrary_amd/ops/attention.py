"""Fused flash attention (hand-written CDNA4 kernels, head_dim 64/128, bf16).

No reference counterpart (the reference's attention is plain TF ops);
this is MI355X-native hot-path work — design notes in docs/kernels.md.

Replaces torch SDPA (AOTriton) on the transformer hot path.  The HIP
kernels (csrc/kernels/attention.hip) keep the whole online softmax in
registers — swapped QK^T so each lane owns one query row, O accumulated
transposed so the rescale is lane-local, P fragments rebuilt with
v_permlane32_swap.  CPU / unsupported shapes fall back to torch SDPA.
"""

import torch
import torch.nn.functional as F

from easyparallellibrary_amd.ops.dispatch import native_ext, use_native
from easyparallellibrary_amd.utils.logging import get_logger

logger = get_logger()
_warned = set()


def _warn_fallback(reason):
    """LOUD (once per reason) SDPA fallback: silent multi-backend
    dispatch on the hot path is exactly what the project brief
    disallows — if a model leaves the native kernels, say so."""
    if reason not in _warned:
        _warned.add(reason)
        logger.warning(
            "native attention unavailable (%s): falling back to torch "
            "SDPA for these shapes", reason)


def _kernel_ok(t):
    return (t.dim() == 4 and t.shape[-1] in (64, 128)
            and t.stride(-1) == 1
            and t.stride(1) % 8 == 0 and t.stride(2) % 8 == 0)


def _drop_params(dropout_p):
    """Quantize the drop probability to 1/256 (the kernel compares
    philox BYTES against an 8-bit threshold): returns (thresh, inv_keep,
    seed).  thresh 0 disables."""
    if not dropout_p:
        return 0, 1.0, 0
    thresh = min(255, max(1, int(round(float(dropout_p) * 256.0))))
    inv_keep = 256.0 / (256 - thresh)
    seed = int(torch.randint(0, 2 ** 62, (1,)).item())
    return thresh, inv_keep, seed


def _mask_tensor(b, h, seq, device):
    mask_w = (seq + 31) // 32
    return torch.empty(b * h * seq * mask_w, dtype=torch.int32,
                       device=device)


class _FlashAttention(torch.autograd.Function):
    """Strided-input flash attention: the kernels consume the qkv-unbind
    views directly (no contiguous() copies); the output is written in
    [B,S,H,D] memory order so the caller's transpose+reshape is free.
    dropout_p > 0 generates the philox keep-mask in-kernel (bit tensor
    republished to the backward kernels; p quantized to 1/256)."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale, dropout_p=0.0):
        if not (_kernel_ok(q) and _kernel_ok(k) and _kernel_ok(v)
                and q.stride() == k.stride() == v.stride()):
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        b, h, seq, d = q.shape
        out = torch.empty(b, seq, h, d, dtype=q.dtype,
                          device=q.device).permute(0, 2, 1, 3)
        lse = torch.empty(b * h * seq, dtype=torch.float32,
                          device=q.device)
        thresh, inv_keep, seed = _drop_params(dropout_p)
        mask = _mask_tensor(b, h, seq, q.device) if thresh else None
        native_ext().attn_fwd(q, k, v, out, lse, scale, causal, mask,
                              seed, thresh, inv_keep)
        if mask is None:
            ctx.save_for_backward(q, k, v, out, lse)
        else:
            ctx.save_for_backward(q, k, v, out, lse, mask)
        ctx.causal = causal
        ctx.scale = scale
        ctx.inv_keep = inv_keep
        return out

    @staticmethod
    def backward(ctx, dout):
        import os
        saved = ctx.saved_tensors  # ONE access: checkpoint unpack hooks
        q, k, v, out, lse = saved[:5]
        mask = saved[5] if len(saved) > 5 else None
        # the bwd kernels re-stage q/k/v/dout tiles every 32-row iteration;
        # strided rows (6 KB apart in the qkv views) measured +405us/call
        # vs 4 contiguization copies at ~43us each — copy for backward only
        # (forward shows no strided penalty)
        orig_q = q
        contig = os.environ.get("EPL_ATTN_BWD_CONTIG", "0") == "1"
        if contig and q.stride(2) != 64:
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        if not _kernel_ok(dout) or (contig and dout.stride(2) != 64):
            dout = dout.contiguous()
        # write grads in the INPUT layout: the qkv-unbind/transpose
        # backward then stacks matching-layout chunks (fast memcpy)
        # instead of strided gathers
        def grad_like(t):
            g = torch.empty_strided(t.shape, t.stride(), dtype=t.dtype,
                                    device=t.device)
            return g
        dq = grad_like(orig_q)
        dk = grad_like(orig_q)
        dv = grad_like(orig_q)
        delta = torch.empty_like(lse)
        # split dK/dV kernels measure ~4% faster than the combined one
        # (3 vs 2 waves/SIMD; A/B in profiles/r01_attention_ab.txt)
        split = os.environ.get("EPL_ATTN_BWD_SPLIT", "1") == "1"
        native_ext().attn_bwd(q, k, v, out, dout, lse, delta, dq, dk, dv,
                              ctx.scale, ctx.causal, split, mask,
                              ctx.inv_keep, None)
        return dq, dk, dv, None, None, None


class _QKVFlashAttention(torch.autograd.Function):
    """Module-level fused path: consumes the [B,S,3,H,D] qkv projection
    buffer directly.  Forward feeds the kernels the q/k/v strided views
    (no unbind copies); backward allocates ONE d_qkv buffer and the
    kernels write dq/dk/dv straight into its slices (they take arbitrary
    shared [B,H,S] stride triplets) — no stack/CatArrayBatchedCopy, no
    3x-oversized ``empty_strided`` transients."""

    @staticmethod
    def forward(ctx, qkv, causal, scale, dropout_p=0.0):
        b, s, three, h, d = qkv.shape
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        out = torch.empty(b, s, h, d, dtype=qkv.dtype,
                          device=qkv.device).permute(0, 2, 1, 3)
        lse = torch.empty(b * h * s, dtype=torch.float32, device=qkv.device)
        thresh, inv_keep, seed = _drop_params(dropout_p)
        mask = _mask_tensor(b, h, s, qkv.device) if thresh else None
        native_ext().attn_fwd(q, k, v, out, lse, scale, causal, mask,
                              seed, thresh, inv_keep)
        if mask is None:
            ctx.save_for_backward(qkv, out, lse)
        else:
            ctx.save_for_backward(qkv, out, lse, mask)
        ctx.causal = causal
        ctx.scale = scale
        ctx.inv_keep = inv_keep
        return out

    @staticmethod
    def backward(ctx, dout):
        saved = ctx.saved_tensors  # ONE access: checkpoint unpack hooks
        qkv, out, lse = saved[:3]
        mask = saved[3] if len(saved) > 3 else None
        q = qkv[:, :, 0].transpose(1, 2)
        k = qkv[:, :, 1].transpose(1, 2)
        v = qkv[:, :, 2].transpose(1, 2)
        if not _kernel_ok(dout):
            dout = dout.contiguous()
        dqkv = torch.empty_like(qkv)
        dq = dqkv[:, :, 0].transpose(1, 2)
        dk = dqkv[:, :, 1].transpose(1, 2)
        dv = dqkv[:, :, 2].transpose(1, 2)
        delta = torch.empty_like(lse)
        native_ext().attn_bwd(q, k, v, out, dout, lse, delta, dq, dk, dv,
                              ctx.scale, ctx.causal, True, mask,
                              ctx.inv_keep, None)
        return dqkv, None, None, None


class _FlashAttentionLse(torch.autograd.Function):
    """Flash attention that ALSO returns the per-row lse (log-sum-exp of
    the scaled scores) as a differentiable output — the building block
    of the native ring-attention path, whose block merge weights depend
    on lse.  The incoming lse gradient folds into the backward's
    per-row delta (dS = P o (dP - (D - dlse)))."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        if not (_kernel_ok(q) and _kernel_ok(k) and _kernel_ok(v)
                and q.stride() == k.stride() == v.stride()):
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        b, h, seq, d = q.shape
        out = torch.empty(b, seq, h, d, dtype=q.dtype,
                          device=q.device).permute(0, 2, 1, 3)
        lse = torch.empty(b * h * seq, dtype=torch.float32,
                          device=q.device)
        native_ext().attn_fwd(q, k, v, out, lse, scale, causal, None,
                              0, 0, 1.0)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        ctx.scale = scale
        return out, lse.view(b, h, seq)

    @staticmethod
    def backward(ctx, dout, dlse):
        q, k, v, out, lse = ctx.saved_tensors
        if not _kernel_ok(dout):
            dout = dout.contiguous()

        def grad_like(t):
            return torch.empty_strided(t.shape, t.stride(), dtype=t.dtype,
                                       device=t.device)
        dq, dk, dv = grad_like(q), grad_like(q), grad_like(q)
        delta = torch.empty_like(lse)
        dlse = (dlse.reshape(-1).contiguous().float()
                if dlse is not None else None)
        native_ext().attn_bwd(q, k, v, out, dout, lse, delta, dq, dk, dv,
                              ctx.scale, ctx.causal, True, None, 1.0,
                              dlse)
        return dq, dk, dv, None, None


def flash_attention_with_lse(q, k, v, causal=False, scale=None):
    """(out, lse) — lse is [B, H, S] fp32 and differentiable.  Native
    kernel on GPU bf16 d 64/128; fp32 torch fallback elsewhere (the
    CPU tests of the ring path run through it)."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if (use_native(q) and q.dtype == torch.bfloat16
            and q.shape[-1] in (64, 128) and q.shape == k.shape):
        # the kernels assume equal q/kv lengths (the ring contract);
        # unequal blocks take the fp32 fallback below
        return _FlashAttentionLse.apply(q, k, v, causal, scale)
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        seq = q.shape[-2]
        maskt = torch.triu(torch.ones(seq, seq, device=q.device,
                                      dtype=torch.bool), 1)
        s = s.masked_fill(maskt, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    out = ((s - lse.unsqueeze(-1)).exp() @ v.float()).to(q.dtype)
    return out, lse


def qkv_flash_attention(qkv, causal=False, scale=None, dropout_p=0.0):
    """qkv: [B, S, 3, H, D] (the fused projection output).  Returns
    attention output as a [B, H, S, D] view whose memory order is
    [B, S, H, D] (the caller's transpose+reshape is free).  Requires the
    native kernels (bf16, D=64); callers fall back to the split-views +
    SDPA path otherwise.  dropout_p is applied in-kernel (philox mask,
    p quantized to 1/256) — pass it only in training mode."""
    if scale is None:
        scale = qkv.shape[-1] ** -0.5
    return _QKVFlashAttention.apply(qkv, causal, scale, dropout_p)


def qkv_native_ok(qkv):
    return (use_native(qkv) and qkv.dtype == torch.bfloat16
            and qkv.dim() == 5 and qkv.shape[-1] in (64, 128)
            and qkv.is_contiguous())


def flash_attention(q, k, v, causal=False, scale=None, allow_native=True,
                    dropout_p=0.0):
    """q,k,v: [B, H, S, D].  Native kernel when bf16/D=64 on GPU; torch
    SDPA otherwise.  dropout_p runs in-kernel on the native path
    (training-mode callers only)."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if allow_native and use_native(q):
        if q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128):
            return _FlashAttention.apply(q, k, v, causal, scale, dropout_p)
        _warn_fallback("dtype={} head_dim={}".format(q.dtype, q.shape[-1]))
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal,
                                          scale=scale, dropout_p=dropout_p)
