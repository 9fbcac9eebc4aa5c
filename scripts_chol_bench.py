"""Microbench: batched Cholesky kernel. Median-of-reps with warmup
(within-process timing discipline; box-to-box single-shot numbers vary 2-5x)."""
import sys, os, time
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from dmosopt_amd import _hipops

dev = torch.device("cuda", 0)
# global warmup
w = torch.randn(512, 512, device=dev); (w @ w).sum().item()

for B, N in [(6, 300), (18, 300), (18, 300), (6, 1000), (2, 2000)]:
    g = torch.Generator().manual_seed(0)
    A = torch.randn(B, N, 16, generator=g)
    K = (A @ A.transpose(1, 2) + 2.0 * torch.eye(N)).float().to(dev).contiguous()
    Kc = K.clone(); logdet, info = _hipops.cholesky_batched_(Kc); torch.cuda.synchronize()
    L_ref = torch.linalg.cholesky(K.double().cpu())
    err = (Kc.double().cpu().tril() - L_ref).abs().max().item()
    times = []
    for rep in range(15):
        Kc = K.clone()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        _hipops.cholesky_batched_(Kc)
        torch.cuda.synchronize(); times.append((time.perf_counter() - t0) * 1e3)
    times = np.array(times[3:])  # drop warm-up reps
    print(f"B={B} N={N}: median {np.median(times):.2f} ms  min {times.min():.2f}  max {times.max():.2f}  err {err:.2e} info={int(info.sum())}")
