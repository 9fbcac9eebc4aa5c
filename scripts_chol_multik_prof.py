"""Profile target: 200 multik cholesky calls at the headline shape (B=12)."""
import os, sys
import torch
torch.set_num_threads(8)
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from dmosopt_amd import _hipops as ext
dev = torch.device("cuda")
B, N = 12, 300
A = torch.randn(B, N, N, device=dev) * 0.1
K = (A @ A.transpose(-1, -2) + 10.0 * torch.eye(N, device=dev)).contiguous()
for _ in range(200):
    W = K.clone()
    ext.cholesky_batched_(W)
torch.cuda.synchronize()
