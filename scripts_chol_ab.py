"""Interleaved A/B of the two diag-factor variants in one process.
DMOSOPT_CHOL_DIAG is read once per process, so this script re-execs itself
... no: the launcher caches mode. Instead we run reps per mode by setting
env BEFORE import in two subprocesses and interleaving at the call level is
impossible; so: run this script twice (wave, lds) back-to-back on the SAME
box within one gpurun call."""
import sys, os, time
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from dmosopt_amd import _hipops

dev = torch.device("cuda", 0)
w = torch.randn(512, 512, device=dev); (w @ w).sum().item()
label = os.environ.get("DMOSOPT_CHOL_DIAG", "wave")
for B, N in [(18, 300), (6, 1000)]:
    g = torch.Generator().manual_seed(0)
    A = torch.randn(B, N, 16, generator=g)
    K = (A @ A.transpose(1, 2) + 2.0 * torch.eye(N)).float().to(dev).contiguous()
    times = []
    for rep in range(15):
        Kc = K.clone()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        _hipops.cholesky_batched_(Kc)
        torch.cuda.synchronize(); times.append((time.perf_counter() - t0) * 1e3)
    t = np.array(times[3:])
    print(f"[{label}] B={B} N={N}: median {np.median(t):.3f} ms min {t.min():.3f}")
