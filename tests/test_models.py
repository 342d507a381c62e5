"""Model zoo sanity on CPU (fused modules fall back to torch ops)."""

import torch
import torch.nn.functional as F

from byteps_amd import models
from byteps_amd.models.bert import BertConfig, BertForPreTraining


def test_resnet50_shapes_and_params():
    m = models.resnet50()
    assert sum(p.numel() for p in m.parameters()) == 25_557_032
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 1000)
    y.sum().backward()
    assert m.conv1.weight.grad is not None


def test_vgg16_params():
    m = models.vgg16()
    assert sum(p.numel() for p in m.parameters()) == 138_357_544
    y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 1000)


def test_bert_large_param_count():
    m = models.bert_large()
    total = sum(p.numel() for p in m.parameters()) / 1e6
    assert 330 < total < 340, total     # reference quotes ≈334M


def test_bert_tiny_trains():
    torch.manual_seed(0)
    m = BertForPreTraining(BertConfig.tiny())
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    ids = torch.randint(0, 1024, (4, 16))
    lab = torch.randint(0, 1024, (4, 16))
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = m.loss(ids, lab)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0]


def test_fused_bn_cpu_fallback_matches_torch():
    from byteps_amd.torch.fused_bn import FusedBNReLU
    torch.manual_seed(1)
    m = FusedBNReLU(16, relu=True)
    bn = torch.nn.BatchNorm2d(16)
    bn.load_state_dict({k: v for k, v in m.state_dict().items()
                        if k in bn.state_dict()}, strict=False)
    x = torch.randn(3, 16, 5, 5)
    y = m(x)
    ref = F.relu(bn(x))
    assert torch.allclose(y, ref, atol=1e-5)


def test_fused_ln_cpu_fallback_matches_torch():
    from byteps_amd.torch.fused_ln import FusedLayerNorm
    m = FusedLayerNorm(32)
    x = torch.randn(4, 7, 32)
    ref = F.layer_norm(x, (32,), m.weight, m.bias, m.eps)
    assert torch.allclose(m(x), ref, atol=1e-6)


def test_mlp():
    m = models.mnist_mlp()
    assert m(torch.randn(2, 1, 28, 28)).shape == (2, 10)
