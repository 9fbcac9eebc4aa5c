import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from dmosopt_amd import _hipops
dev = torch.device("cuda", 0)
g = torch.Generator().manual_seed(0)
A = torch.randn(18, 300, 16, generator=g)
K = (A @ A.transpose(1, 2) + 2.0 * torch.eye(300)).float().to(dev).contiguous()
for _ in range(5):
    Kc = K.clone(); _hipops.cholesky_batched_(Kc)
torch.cuda.synchronize()
print("done")
