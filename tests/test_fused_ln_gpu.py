"""Fused LayerNorm gfx950 kernels vs plain PyTorch fp32 reference."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from byteps_amd.torch.fused_ln import FusedLayerNorm  # noqa: E402


@pytest.mark.parametrize("shape", [(8, 128, 1024), (4, 7, 2048),
                                   (64, 512), (3, 5, 64), (1000, 256)])
def test_fused_ln_forward_backward(shape):
    C = shape[-1]
    g = torch.Generator().manual_seed(C)
    x = torch.randn(shape, generator=g).to("cuda", torch.bfloat16)
    x_t = x.detach().clone().requires_grad_(True)
    x_ref = x.detach().float().clone().requires_grad_(True)

    m = FusedLayerNorm(C, eps=1e-12).cuda()
    with torch.no_grad():
        m.weight.add_(torch.randn_like(m.weight) * 0.3)
        m.bias.add_(torch.randn_like(m.bias) * 0.2)
    w_ref = m.weight.detach().float().clone().requires_grad_(True)
    b_ref = m.bias.detach().float().clone().requires_grad_(True)

    y = m(x_t)
    assert y.dtype == torch.bfloat16
    y_ref = F.layer_norm(x_ref, (C,), w_ref, b_ref, 1e-12)
    assert torch.allclose(y.float(), y_ref, atol=3e-2, rtol=3e-2)

    g16 = torch.randn(shape, generator=g).to("cuda", torch.bfloat16)
    y.backward(g16)
    y_ref.backward(g16.float())
    torch.cuda.synchronize()
    assert torch.allclose(x_t.grad.float(), x_ref.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(m.weight.grad, w_ref.grad, atol=2e-1, rtol=2e-2)
    assert torch.allclose(m.bias.grad, b_ref.grad, atol=2e-1, rtol=2e-2)


def test_fused_ln_unsupported_falls_back():
    m = FusedLayerNorm(96).cuda()     # cpt=12: not a power of two
    x = torch.randn(4, 96, device="cuda", dtype=torch.bfloat16)
    y = m(x)
    ref = F.layer_norm(x.float(), (96,), m.weight, m.bias, m.eps)
    assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2)


def test_bert_block_with_fused_ln_trains():
    from byteps_amd.models.bert import BertConfig, BertForPreTraining
    cfg = BertConfig(vocab_size=2048, hidden=512, layers=2, heads=8,
                     intermediate=1024, max_pos=64)
    torch.manual_seed(0)
    m = BertForPreTraining(cfg).cuda()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    ids = torch.randint(0, 2048, (4, 32), device="cuda")
    lab = torch.randint(0, 2048, (4, 32), device="cuda")
    losses = []
    for _ in range(10):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = m.loss(ids, lab)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert losses[-1] < losses[0], losses
