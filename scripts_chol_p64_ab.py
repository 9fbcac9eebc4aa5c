"""Same-box A/B: PANEL64 vs PANEL32 multik Cholesky + large-N regime check.

Run twice in one gpurun call (the launcher caches the env choice):
  DMOSOPT_CHOL_PANEL=32 python scripts_chol_p64_ab.py
  DMOSOPT_CHOL_PANEL=64 python scripts_chol_p64_ab.py
Also re-checks the native-vs-rocSOLVER crossover at
N in {1024, 2048, 4096} with direct calls (bypassing the dispatch).
"""

import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch

from dmosopt_amd import _hipops

dev = torch.device("cuda", 0)
w = torch.randn(512, 512, device=dev)
(w @ w).sum().item()
label = os.environ.get("DMOSOPT_CHOL_PANEL", "64")


def make_k(B, N, seed=0):
    g = torch.Generator().manual_seed(seed)
    A = torch.randn(B, N, 16, generator=g)
    return (A @ A.transpose(1, 2) / 16 + 2.0 * torch.eye(N)).float().to(dev).contiguous()


# correctness first (both panel modes must match the fp64 factor)
for B, N in [(12, 300), (4, 513)]:
    K = make_k(B, N, 1)
    Kc = K.clone()
    logdet, info = _hipops.cholesky_batched_(Kc)
    assert int(info.abs().sum()) == 0
    L_ref = torch.linalg.cholesky(K.double().cpu())
    err = (Kc.double().cpu().tril() - L_ref).abs().max()
    ld_ref = torch.log(torch.diagonal(L_ref, dim1=1, dim2=2)).sum(dim=1)
    ld_err = (logdet.double().cpu() - ld_ref).abs().max()
    print(f"[{label}] correctness B={B} N={N}: maxerr {float(err):.2e} logdet {float(ld_err):.2e}")
    assert float(err) < 5e-3 and float(ld_err) < 1e-3

for B, N in [(12, 300), (36, 300), (12, 513), (6, 1000)]:
    K = make_k(B, N)
    times = []
    for rep in range(15):
        Kc = K.clone()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        _hipops.cholesky_batched_(Kc)
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) * 1e3)
    t = np.array(times[3:])
    print(f"[panel{label}] B={B} N={N}: median {np.median(t):.3f} ms min {t.min():.3f}")

# large-N regime (only meaningful once per box; runs under both labels)
print("--- large-N native multik vs torch/rocSOLVER ---")
for B, N in [(4, 1024), (2, 2048), (1, 4096)]:
    K = make_k(B, N)
    for name, fn in (
        ("native", lambda Kc: _hipops.cholesky_batched_(Kc)),
        ("torch", lambda Kc: torch.linalg.cholesky_ex(Kc)),
    ):
        times = []
        for rep in range(7):
            Kc = K.clone()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            fn(Kc)
            torch.cuda.synchronize()
            times.append((time.perf_counter() - t0) * 1e3)
        t = np.array(times[2:])
        print(f"[{label}] {name} B={B} N={N}: median {np.median(t):.3f} ms")
