"""Numerics of the hand-written CDNA4 kernels vs plain fp32 torch
references (runs on an MI355X via gpurun)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from easyparallellibrary_amd import _C
else:
    _C = None


def dev():
    return torch.device("cuda:0")


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    n = 12345
    master = torch.randn(n, device=dev())
    grad = torch.randn(n, device=dev(), dtype=torch.bfloat16)
    m = torch.zeros(n, device=dev())
    v = torch.zeros(n, device=dev())
    pbf = master.to(torch.bfloat16)

    ref_p = master.clone()
    ref_m, ref_v = m.clone(), v.clone()
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.01
    for step in (1, 2, 3):
        _C.fused_adamw(master, pbf, grad, m, v, lr, b1, b2, eps, wd, step,
                       1.0)
        gf = grad.float()
        ref_m.mul_(b1).add_(gf, alpha=1 - b1)
        ref_v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
        mh = ref_m / (1 - b1 ** step)
        vh = ref_v / (1 - b2 ** step)
        ref_p.add_(mh / (vh.sqrt() + eps) + wd * ref_p, alpha=-lr)
    torch.cuda.synchronize()
    assert torch.allclose(master, ref_p, atol=1e-6, rtol=1e-5)
    assert torch.allclose(m, ref_m, atol=1e-6, rtol=1e-5)
    assert torch.allclose(v, ref_v, atol=1e-6, rtol=1e-5)
    assert torch.allclose(pbf.float(), ref_p, atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("dtype,cols,rows", [
    (torch.bfloat16, 1024, 512), (torch.float32, 1024, 512),
    (torch.bfloat16, 512, 512), (torch.bfloat16, 1600, 512),
    (torch.bfloat16, 2048, 512), (torch.bfloat16, 4096, 512),
    (torch.bfloat16, 1024, 3), (torch.bfloat16, 1024, 8191)])
def test_layer_norm_fwd_bwd(dtype, cols, rows):
    torch.manual_seed(1)
    x = torch.randn(rows, cols, device=dev(), dtype=dtype)
    gamma = torch.randn(cols, device=dev(), dtype=dtype)
    beta = torch.randn(cols, device=dev(), dtype=dtype)
    dy = torch.randn(rows, cols, device=dev(), dtype=dtype)

    mean = torch.empty(rows, dtype=torch.float32, device=dev())
    rstd = torch.empty(rows, dtype=torch.float32, device=dev())
    out = torch.empty_like(x)
    _C.layer_norm_fwd(out, x, None, None, gamma, beta, mean, rstd, 1e-5)

    xf = x.float()
    ref_mu = xf.mean(dim=1)
    ref_rstd = (xf.var(dim=1, unbiased=False) + 1e-5).rsqrt()
    xhat = (xf - ref_mu[:, None]) * ref_rstd[:, None]
    ref_out = xhat * gamma.float() + beta.float()
    tol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    torch.cuda.synchronize()
    assert torch.allclose(mean, ref_mu, atol=1e-4, rtol=1e-4)
    assert torch.allclose(out.float(), ref_out, atol=tol, rtol=tol)

    dx = torch.empty_like(x)
    dgamma = torch.zeros(cols, dtype=torch.float32, device=dev())
    dbeta = torch.zeros(cols, dtype=torch.float32, device=dev())
    _C.layer_norm_bwd(dx, dgamma, dbeta, dy, x, gamma, mean, rstd)

    xr = xf.clone().requires_grad_(True)
    gr = gamma.float().clone().requires_grad_(True)
    br = beta.float().clone().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xr, (cols,), gr, br, 1e-5)
    ref.backward(dy.float())
    torch.cuda.synchronize()
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(dx.float(), xr.grad, atol=tol, rtol=tol)
    assert torch.allclose(dgamma, gr.grad, atol=0.5, rtol=1e-2)
    assert torch.allclose(dbeta, br.grad, atol=0.5, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_bias_gelu(dtype):
    torch.manual_seed(2)
    rows, cols = 256, 512
    x = torch.randn(rows, cols, device=dev(), dtype=dtype)
    bias = torch.randn(cols, device=dev(), dtype=dtype)
    dy = torch.randn(rows, cols, device=dev(), dtype=dtype)
    out = torch.empty_like(x)
    _C.bias_gelu_fwd(out, x, bias)

    xb = (x.float() + bias.float()).requires_grad_(False)
    ref = 0.5 * xb * (1 + torch.tanh(0.7978845608 * (xb + 0.044715 * xb**3)))
    tol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    torch.cuda.synchronize()
    assert torch.allclose(out.float(), ref, atol=tol, rtol=tol)

    dx = torch.empty_like(x)
    dbias = torch.zeros(cols, dtype=torch.float32, device=dev())
    _C.bias_gelu_bwd(dx, dbias, dy, x, bias)
    xr = x.float().clone().requires_grad_(True)
    br = bias.float().clone().requires_grad_(True)
    xb2 = xr + br
    ref2 = 0.5 * xb2 * (1 + torch.tanh(0.7978845608 *
                                       (xb2 + 0.044715 * xb2**3)))
    ref2.backward(dy.float())
    torch.cuda.synchronize()
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(dx.float(), xr.grad, atol=tol, rtol=tol)
    assert torch.allclose(dbias, br.grad, atol=0.5, rtol=1e-2)


def test_cross_entropy_full_vocab():
    torch.manual_seed(3)
    from easyparallellibrary_amd.env import Env  # noqa: F401  (config)
    import easyparallellibrary_amd as epl
    epl.init()
    from easyparallellibrary_amd.ops.distributed_losses import (
        vocab_parallel_cross_entropy)
    rows, vocab = 128, 1000
    logits = torch.randn(rows, vocab, device=dev(), dtype=torch.bfloat16,
                         requires_grad=True)
    targets = torch.randint(0, vocab, (rows,), device=dev())
    targets[::7] = -100
    losses = vocab_parallel_cross_entropy(logits, targets)
    loss = losses.sum() / (targets != -100).sum()
    loss.backward()

    lref = logits.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lref, targets,
                                            ignore_index=-100)
    ref.backward()
    torch.cuda.synchronize()
    assert abs(loss.item() - ref.item()) < 2e-2
    assert torch.allclose(logits.grad.float(), lref.grad, atol=2e-3,
                          rtol=5e-2)


def test_scale_and_casts():
    torch.manual_seed(4)
    x = torch.randn(1001, device=dev())
    ref = x * 0.5
    _C.scale_(x, 0.5)
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)

    src = torch.randn(777, device=dev())
    dst = torch.empty(777, device=dev(), dtype=torch.bfloat16)
    _C.f32_to_bf16(dst, src)
    back = torch.empty(777, device=dev())
    _C.bf16_to_f32(back, dst)
    torch.cuda.synchronize()
    assert torch.allclose(back, src.to(torch.bfloat16).float())

    out = torch.zeros(1, device=dev())
    _C.sqnorm(src, out)
    torch.cuda.synchronize()
    assert torch.allclose(out, (src * src).sum().reshape(1), rtol=1e-4)


def test_rccl_single_rank_comm():
    from easyparallellibrary_amd.comm.backend import RcclCommunicator
    c = RcclCommunicator("t_gpu1", [0])
    t = torch.randn(1024, device=dev())
    ref = t.clone()
    c.all_reduce(t)
    c.synchronize()
    assert torch.allclose(t, ref)
    c.destroy()


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_fused_residual_layer_norm_gpu(dtype):
    torch.manual_seed(9)
    from easyparallellibrary_amd.ops.layer_norm import FusedLayerNorm
    import easyparallellibrary_amd as epl
    epl.init()
    ln = FusedLayerNorm(1024).to(dev(), dtype)
    x = torch.randn(256, 1024, device=dev(), dtype=dtype, requires_grad=True)
    r = torch.randn(256, 1024, device=dev(), dtype=dtype, requires_grad=True)
    y = ln(x, residual=r)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().clone().requires_grad_(True)
    rf = r.detach().float().clone().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(
        xf + rf, (1024,), ln.weight.detach().float(),
        ln.bias.detach().float())
    yr.backward(dy.float())
    torch.cuda.synchronize()
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(y.float(), yr, atol=tol, rtol=tol)
    assert torch.allclose(x.grad.float(), xf.grad, atol=tol, rtol=tol)
    assert torch.allclose(r.grad.float(), rf.grad, atol=tol, rtol=tol)


@pytest.mark.parametrize("cols,dtype", [(1024, torch.bfloat16),
                                        (3072, torch.bfloat16),
                                        (1024, torch.float32)])
def test_colsum(cols, dtype):
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.bias_linear import fused_colsum
    epl.init()
    torch.manual_seed(8)
    dy = torch.randn(4096, cols, device="cuda", dtype=dtype)
    got = fused_colsum(dy)
    want = dy.float().sum(dim=0)
    torch.cuda.synchronize()
    err = (got.float() - want).abs().max().item()
    rel = err / want.abs().max().clamp_min(1e-6).item()
    assert rel < 2e-2, (err, rel)


def test_fused_bias_linear_grads():
    import easyparallellibrary_amd as epl
    from easyparallellibrary_amd.ops.bias_linear import FusedBiasLinear
    epl.init()
    torch.manual_seed(9)
    lin = FusedBiasLinear(64, 128).cuda().to(torch.bfloat16)
    ref = torch.nn.Linear(64, 128).cuda().to(torch.bfloat16)
    ref.load_state_dict(lin.state_dict())
    x = torch.randn(8, 16, 64, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    xr = x.detach().clone().requires_grad_(True)
    y = lin(x); yr = ref(xr)
    assert torch.equal(y, yr)
    dy = torch.randn_like(y)
    y.backward(dy); yr.backward(dy)
    torch.cuda.synchronize()
    for got, want, name in ((x.grad, xr.grad, "dx"),
                            (lin.weight.grad, ref.weight.grad, "dw"),
                            (lin.bias.grad, ref.bias.grad, "db")):
        err = (got.float() - want.float()).abs().max().item()
        rel = err / want.float().abs().max().clamp_min(1e-6).item()
        assert rel < 3e-2, (name, err, rel)
