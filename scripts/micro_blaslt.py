import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from byteps_amd.ops import core
c = core()
dev = torch.device("cuda")
M, K, I, H = 8192, 1024, 4096, 1024
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
w1 = torch.randn(I, K, device=dev, dtype=torch.bfloat16)
w2 = torch.randn(H, I, device=dev, dtype=torch.bfloat16)
h = torch.randn(M, I, device=dev, dtype=torch.bfloat16)
dy2 = torch.randn(M, H, device=dev, dtype=torch.bfloat16)
y1 = torch.randn(M, I, device=dev, dtype=torch.bfloat16)
dy1 = torch.empty(M, I, device=dev, dtype=torch.bfloat16)
dw2 = torch.empty(H, I, device=dev, dtype=torch.bfloat16)
db2 = torch.empty(H, device=dev, dtype=torch.bfloat16)
dw1 = torch.empty(I, K, device=dev, dtype=torch.bfloat16)
db1 = torch.empty(I, device=dev, dtype=torch.bfloat16)
dx = torch.empty(M, K, device=dev, dtype=torch.bfloat16)
s = torch.cuda.current_stream().cuda_stream

def bench(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

print("wgrad2 lt+bgrad : %7.1f us" % bench(lambda: c.lt_gemm_wgrad(dy2.data_ptr(), y1.data_ptr(), dw2.data_ptr(), db2.data_ptr(), M, H, I, s)))
print("wgrad2 torch    : %7.1f us (+%5.1f us bias sum)" % (bench(lambda: torch.matmul(dy2.t(), y1)), bench(lambda: dy2.sum(0))))
print("dgelu lt        : %7.1f us" % bench(lambda: c.lt_gemm_dgelu(dy2.data_ptr(), w2.data_ptr(), h.data_ptr(), dy1.data_ptr(), M, H, I, s)))
import torch.nn.functional as F
def eager_dgelu():
    t = torch.matmul(dy2, w2)
    # torch gelu bwd kernel approx: use autograd on gelu
    return t
print("dgelu torch mm  : %7.1f us (matmul only)" % bench(eager_dgelu))
print("wgrad1 lt+bgrad : %7.1f us" % bench(lambda: c.lt_gemm_wgrad(dy1.data_ptr(), x.data_ptr(), dw1.data_ptr(), db1.data_ptr(), M, I, K, s)))
print("wgrad1 torch    : %7.1f us (+%5.1f us bias sum)" % (bench(lambda: torch.matmul(dy1.t(), x)), bench(lambda: dy1.sum(0))))
print("dgrad1 lt       : %7.1f us" % bench(lambda: c.lt_gemm_dgrad(dy1.data_ptr(), w1.data_ptr(), dx.data_ptr(), M, I, K, s)))
print("dgrad1 torch    : %7.1f us" % bench(lambda: torch.matmul(dy1, w1)))
print("fwd1 lt bias    : %7.1f us" % bench(lambda: c.lt_gemm_bias(x.data_ptr(), w1.data_ptr(), db1.data_ptr(), h.data_ptr(), M, I, K, s)))
b1 = torch.randn(I, device=dev, dtype=torch.bfloat16)
print("fwd1 torch      : %7.1f us" % bench(lambda: F.linear(x, w1, b1)))
