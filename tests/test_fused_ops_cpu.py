"""CPU-path numerics of the fused-op wrappers (autograd correctness of the
fused residual LayerNorm and bias-GeLU against plain torch)."""

import torch

import easyparallellibrary_amd as epl
from easyparallellibrary_amd.ops.bias_gelu import FusedBiasGelu
from easyparallellibrary_amd.ops.layer_norm import FusedLayerNorm


def setup_module(_):
    epl.init()


def test_layer_norm_matches_torch():
    torch.manual_seed(0)
    ln = FusedLayerNorm(64)
    ref = torch.nn.LayerNorm(64)
    with torch.no_grad():
        ref.weight.copy_(ln.weight)
        ref.bias.copy_(ln.bias)
    x = torch.randn(8, 64, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y = ln(x)
    yr = ref(x2)
    assert torch.allclose(y, yr, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(ln.weight.grad, ref.weight.grad, atol=1e-5)
    assert torch.allclose(ln.bias.grad, ref.bias.grad, atol=1e-5)


def test_fused_residual_layer_norm():
    torch.manual_seed(1)
    ln = FusedLayerNorm(32)
    x = torch.randn(4, 32, requires_grad=True)
    r = torch.randn(4, 32, requires_grad=True)
    y = ln(x, residual=r)

    x2 = x.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(x2 + r2, (32,), ln.weight.detach(),
                                        ln.bias.detach())
    assert torch.allclose(y, yr, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(r.grad, r2.grad, atol=1e-5)


def test_fused_ln_with_sum_stream():
    torch.manual_seed(2)
    ln = FusedLayerNorm(16)
    x = torch.randn(3, 16, requires_grad=True)
    r = torch.randn(3, 16, requires_grad=True)
    y, s = ln.forward_with_sum(x, r)
    loss = y.sum() + (s * 2).sum()

    x2 = x.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    s2 = x2 + r2
    y2 = torch.nn.functional.layer_norm(s2, (16,), ln.weight.detach(),
                                        ln.bias.detach())
    loss2 = y2.sum() + (s2 * 2).sum()
    loss.backward()
    loss2.backward()
    assert torch.allclose(loss, loss2, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(r.grad, r2.grad, atol=1e-5)


def test_bias_gelu_matches_torch():
    torch.manual_seed(3)
    bg = FusedBiasGelu(24)
    with torch.no_grad():
        bg.bias.uniform_(-1, 1)
    x = torch.randn(5, 24, requires_grad=True)
    y = bg(x)
    x2 = x.detach().clone().requires_grad_(True)
    yr = torch.nn.functional.gelu(x2 + bg.bias.detach(), approximate="tanh")
    assert torch.allclose(y, yr, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)


def test_lamb_cpu_matches_reference_math():
    """Eager LAMB path vs a hand-computed single step on one tensor."""
    import torch.nn as nn
    import easyparallellibrary_amd as epl
    epl.init()
    torch.manual_seed(7)
    with epl.replicate(1):
        model = nn.Linear(16, 16, bias=False)
    engine = epl.Engine(model, loss_fn=nn.MSELoss(), optimizer="lamb",
                        lr=1e-2)
    w0 = model.weight.detach().clone()
    x, y = torch.randn(8, 16), torch.randn(8, 16)
    # manual reference: same forward/backward, LAMB update by hand
    wm = w0.clone().requires_grad_(True)
    loss = nn.MSELoss()(x @ wm.t(), y)
    loss.backward()
    g = wm.grad.detach()
    b1, b2, eps, wd, lr = 0.9, 0.999, 1e-6, 0.01, 1e-2
    m = (1 - b1) * g
    v = (1 - b2) * g * g
    update = (m / (1 - b1)) / ((v / (1 - b2)).sqrt() + eps) + wd * w0
    trust = w0.norm() / update.norm().clamp_min(1e-12)
    trust = trust.clamp(max=10.0)
    expect = w0 - lr * trust * update

    engine.train_step(x, y)
    got = model.weight.detach()
    # per-chunk trust ratios may differ from the whole-tensor one if the
    # implementation chunks differently; require close agreement
    assert (got - expect).abs().max() < 5e-3, (
        (got - expect).abs().max(), trust)
