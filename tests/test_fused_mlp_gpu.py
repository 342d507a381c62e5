"""hipBLASLt epilogue-fused BERT MLP block vs an fp32 eager reference
(same tanh-GELU).  bf16 GEMMs with fp32 accumulate → tolerances are
bf16-scale."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ref(x, w1, b1, w2, b2):
    h = torch.nn.functional.gelu(x @ w1.t() + b1, approximate="tanh")
    return h @ w2.t() + b2, h


@pytest.mark.parametrize("M,K,I", [(256, 512, 2048), (1024, 1024, 4096)])
def test_fused_mlp_forward_backward(M, K, I):
    from byteps_amd.torch.fused_mlp import fused_mlp, fused_mlp_available
    assert fused_mlp_available()
    H = K
    torch.manual_seed(0)
    dev = torch.device("cuda")
    x = torch.randn(M, K, device=dev) * 0.5
    w1 = torch.randn(I, K, device=dev) * 0.02
    b1 = torch.randn(I, device=dev) * 0.1
    w2 = torch.randn(H, I, device=dev) * 0.02
    b2 = torch.randn(H, device=dev) * 0.1
    g = torch.randn(M, H, device=dev) * 0.1

    # fp32 eager reference
    xr = x.clone().requires_grad_()
    w1r = w1.clone().requires_grad_()
    b1r = b1.clone().requires_grad_()
    w2r = w2.clone().requires_grad_()
    b2r = b2.clone().requires_grad_()
    yr, _h = _ref(xr, w1r, b1r, w2r, b2r)
    yr.backward(g)

    # fused path (casts to bf16 internally)
    xf = x.clone().requires_grad_()
    w1f = w1.clone().requires_grad_()
    b1f = b1.clone().requires_grad_()
    w2f = w2.clone().requires_grad_()
    b2f = b2.clone().requires_grad_()
    yf = fused_mlp(xf, w1f, b1f, w2f, b2f)
    yf.backward(g.to(yf.dtype))
    torch.cuda.synchronize()

    def close(a, b, tag, rtol=6e-2):
        a = a.float()
        b = b.float()
        denom = b.abs().mean().clamp_min(1e-6)
        err = (a - b).abs().mean() / denom
        assert err < rtol, "%s rel err %.4f" % (tag, float(err))

    close(yf, yr, "forward")
    close(xf.grad, xr.grad, "dx")
    close(w1f.grad, w1r.grad, "dw1")
    close(b1f.grad, b1r.grad, "db1")
    close(w2f.grad, w2r.grad, "dw2")
    close(b2f.grad, b2r.grad, "db2")


def test_bert_layer_fused_vs_eager_numerics():
    """Whole encoder layer: fused and eager MLP paths must agree to bf16
    tolerance on the same weights."""
    from byteps_amd.models import bert as B
    torch.manual_seed(1)
    cfg = B.BertConfig(hidden=256, layers=1, heads=4, intermediate=1024)
    layer = B.EncoderLayer(cfg).cuda()
    x = torch.randn(4, 32, 256, device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        layer._fused_mlp = True
        y_fused = layer(x).float()
        layer._fused_mlp = False
        y_eager = layer(x).float()
    err = (y_fused - y_eager).abs().mean() / y_eager.abs().mean()
    assert err < 3e-2, "fused layer diverges: %.4f" % float(err)
