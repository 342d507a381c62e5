"""Fused BN(+residual)(+ReLU) gfx950 kernels vs plain PyTorch fp32
reference — forward, backward, running stats."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from byteps_amd.torch.fused_bn import FusedBNReLU  # noqa: E402


def _mk(N, C, H, W, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(N, C, H, W, generator=g).to("cuda", torch.bfloat16)
    return x.to(memory_format=torch.channels_last)


@pytest.mark.parametrize("shape", [(8, 64, 32, 32), (4, 256, 14, 14),
                                   (2, 2048, 7, 7), (3, 40, 9, 11)])
@pytest.mark.parametrize("relu", [True, False])
def test_fused_bn_forward_backward(shape, relu):
    N, C, H, W = shape
    x = _mk(N, C, H, W, seed=C)
    x_ref = x.detach().float().clone().requires_grad_(True)
    x_t = x.detach().clone().requires_grad_(True)

    m = FusedBNReLU(C, relu=relu).cuda().train()
    # fp32 reference on the same init
    ref_w = m.weight.detach().float().clone().requires_grad_(True)
    ref_b = m.bias.detach().float().clone().requires_grad_(True)
    rm, rv = torch.zeros(C, device="cuda"), torch.ones(C, device="cuda")

    y = m(x_t)
    y_ref = F.batch_norm(x_ref, rm, rv, ref_w, ref_b, True, 0.1, m.eps)
    if relu:
        y_ref = F.relu(y_ref)

    tol = 3e-2  # bf16 storage
    assert torch.allclose(y.float(), y_ref, atol=tol, rtol=tol)
    assert torch.allclose(m.running_mean, rm, atol=1e-3, rtol=1e-2)
    assert torch.allclose(m.running_var, rv, atol=1e-2, rtol=2e-2)

    # give both sides the SAME bf16-rounded upstream gradient — otherwise
    # the channel reductions accumulate the quantization difference over
    # M elements and dbeta/dgamma drift apart at large M
    g16 = torch.randn_like(y_ref).to(torch.bfloat16)
    y.backward(g16.to(memory_format=torch.channels_last))
    y_ref.backward(g16.float())
    torch.cuda.synchronize()
    assert torch.allclose(x_t.grad.float(), x_ref.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(m.weight.grad, ref_w.grad, atol=2e-1, rtol=2e-2)
    assert torch.allclose(m.bias.grad, ref_b.grad, atol=2e-1, rtol=2e-2)


def test_fused_bn_residual():
    N, C, H, W = 4, 128, 16, 16
    x = _mk(N, C, H, W, 1)
    res = _mk(N, C, H, W, 2)
    x_t = x.detach().clone().requires_grad_(True)
    r_t = res.detach().clone().requires_grad_(True)
    x_ref = x.detach().float().clone().requires_grad_(True)
    r_ref = res.detach().float().clone().requires_grad_(True)

    m = FusedBNReLU(C, relu=True).cuda().train()
    with torch.no_grad():
        m.weight.mul_(0).add_(torch.rand_like(m.weight) + 0.5)
        m.bias.add_(torch.randn_like(m.bias) * 0.1)
    ref_w = m.weight.detach().float().clone().requires_grad_(True)
    ref_b = m.bias.detach().float().clone().requires_grad_(True)
    rm, rv = torch.zeros(C, device="cuda"), torch.ones(C, device="cuda")

    y = m(x_t, residual=r_t)
    y_ref = F.relu(
        F.batch_norm(x_ref, rm, rv, ref_w, ref_b, True, 0.1, m.eps) + r_ref)
    assert torch.allclose(y.float(), y_ref, atol=3e-2, rtol=3e-2)

    g16 = torch.randn_like(y_ref).to(torch.bfloat16)
    y.backward(g16.to(memory_format=torch.channels_last))
    y_ref.backward(g16.float())
    torch.cuda.synchronize()
    assert torch.allclose(x_t.grad.float(), x_ref.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(r_t.grad.float(), r_ref.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(m.weight.grad, ref_w.grad, atol=2e-1, rtol=2e-2)


def test_fused_bn_eval_mode():
    C = 64
    x = _mk(2, C, 8, 8, 3)
    m = FusedBNReLU(C, relu=False).cuda()
    with torch.no_grad():
        m.running_mean.add_(torch.randn(C, device="cuda") * 0.3)
        m.running_var.mul_(torch.rand(C, device="cuda") + 0.5)
    m.eval()
    with torch.no_grad():
        y = m(x)
        y_ref = F.batch_norm(x.float(), m.running_mean, m.running_var,
                             m.weight, m.bias, False, 0.1, m.eps)
    assert torch.allclose(y.float(), y_ref, atol=3e-2, rtol=3e-2)


def test_resnet50_uses_fused_bn_and_trains():
    from byteps_amd.models import resnet50
    torch.manual_seed(0)
    net = resnet50().cuda().to(memory_format=torch.channels_last)
    x = torch.randn(4, 3, 64, 64, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (4,), device="cuda")
    opt = torch.optim.SGD(net.parameters(), lr=0.05, momentum=0.9)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(net(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert losses[-1] < losses[0], losses
