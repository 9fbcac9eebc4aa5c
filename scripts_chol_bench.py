"""Microbench: batched Cholesky kernel vs torch (rocSOLVER)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from dmosopt_amd import _hipops

dev = torch.device("cuda", 0)

# probe: rocSOLVER cholesky alone, before any custom kernel
K0 = (torch.randn(2, 64, 8) @ torch.randn(2, 8, 64)).float()
K0 = (K0 @ K0.transpose(1,2) + 50*torch.eye(64)).to(dev)
try:
    L, info = torch.linalg.cholesky_ex(K0)
    torch.cuda.synchronize()
    print("rocSOLVER probe OK")
except Exception as e:
    print("rocSOLVER probe FAILED:", e)

for B, N in [(6, 300), (6, 301), (18, 300), (6, 1000), (2, 2000), (64, 300)]:
    g = torch.Generator().manual_seed(0)
    A = torch.randn(B, N, 16, generator=g)
    K = (A @ A.transpose(1, 2) + 2.0 * torch.eye(N)).float().to(dev).contiguous()
    Kc = K.clone(); logdet, info = _hipops.cholesky_batched_(Kc); torch.cuda.synchronize()
    L_ref = torch.linalg.cholesky(K.double().cpu())
    err = (Kc.double().cpu().tril() - L_ref).abs().max().item()
    ld_ref = torch.log(torch.diagonal(L_ref, dim1=1, dim2=2)).sum(1)
    ld_err = (logdet.double().cpu() - ld_ref).abs().max().item()
    reps = 20
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps):
        Kc = K.clone(); _hipops.cholesky_batched_(Kc)
    torch.cuda.synchronize(); t1 = time.perf_counter()
    ours = (t1 - t0) / reps * 1e3
    try:
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(reps):
            torch.linalg.cholesky_ex(K)
        torch.cuda.synchronize(); t1 = time.perf_counter()
        rocsolver = (t1 - t0) / reps * 1e3
    except Exception as e:
        rocsolver = float("nan"); print("   rocSOLVER failed:", type(e).__name__)
    print(f"B={B} N={N}: ours {ours:.2f} ms  rocSOLVER {rocsolver:.2f} ms  maxerr {err:.2e} lderr {ld_err:.2e} info={int(info.sum())}")
