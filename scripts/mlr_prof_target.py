"""rocprof target: 20x rocBLAS pair then 20x MFMA rb0 (kernel stats A/B)."""
import torch

from harmony_amd import ops

torch.manual_seed(0)
X = torch.randn(16384, 16384, device="cuda")
W = torch.randn(10, 16384, device="cuda") * 0.01
y = torch.randint(0, 10, (16384,), device="cuda")
wt = torch.zeros((16384, 16), device="cuda")
for _ in range(20):
    p, l, c = ops.mlr_forward(X, W, y)
    g = ops.mlr_grad_gemm(p, X)
torch.cuda.synchronize()
for _ in range(20):
    ops.mlr_step_mfma(X, W, y, row_block=0, Wt_buf=wt)
torch.cuda.synchronize()
print("done")
