import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_distributed_train_amd.ops.backend import ext
torch.cuda.init()
dev = "cuda"
# 1. plain gemm_nt (the test that crashed)
A = torch.randn(128, 64, device=dev, dtype=torch.bfloat16)
B = torch.randn(96, 64, device=dev, dtype=torch.bfloat16)
C = ext().gemm_nt(A, B, None, False)
torch.cuda.synchronize(); print("gemm_nt ok", C.shape)
# 2. conv fwd with stats
x = torch.randn(2, 30, 30, 64, device=dev, dtype=torch.bfloat16)
w = torch.randn(128, 64, 3, 3, device=dev) * 0.05
wt = w.permute(0, 2, 3, 1).to(torch.bfloat16).contiguous()
y, part = ext().conv2d_fwd(x, wt, 1, 1, True)
torch.cuda.synchronize(); print("conv stats ok")
# 3. bn premask
mean, var = ext().bn_stats(y)
invstd = (var + 1e-5).rsqrt()
wc = torch.ones(128, device=dev); bc = torch.zeros(128, device=dev)
dy = torch.randn_like(y)
s2, sx2 = ext().bn_bwd_reduce(y, dy, None, mean, invstd=invstd, weight=wc, bias=bc)
torch.cuda.synchronize(); print("bn premask reduce ok")
dx = ext().bn_bwd_elemt(y, dy, None, mean, invstd, wc, s2, sx2, float(y.numel()//128), False, bias=bc)[0]
torch.cuda.synchronize(); print("bn premask elemt ok", float(dx.abs().mean()))
