import sys, os
sys.path.insert(0, "/root/repo")
import numpy as np, torch
from dmosopt_amd import ops, _hipops
dev = torch.device("cuda", 0)
g = torch.Generator().manual_seed(0)
Xq = torch.rand(512, 30, generator=g).float().to(dev)
X = torch.rand(300, 30, generator=g).float().to(dev)
theta = torch.tensor([[0.0, -0.7, -9.0], [0.3, 0.1, -8.0]]).float().to(dev)
for _ in range(50):
    K = ops.matern_cross_bf16_kernel(Xq, X, theta, 2.5, False)
A = torch.randn(4, 300, 40, generator=g)
Kc = (A @ A.transpose(1, 2) / 40 + 0.5 * torch.eye(300)).float().to(dev).contiguous()
for _ in range(20):
    ops.chol_factor_batched_bf16(Kc.clone())
torch.cuda.synchronize()
print("done")
