"""Median per-call latency of the batched triangular-solve kernels.

Shapes mirror the headline epoch: the SCE-UA MLL search solves B candidate
systems (N=300, R=1) per stage; GP predict does one forward solve with
R = query-batch columns.
"""
import os, sys, time

import torch

torch.set_num_threads(min(8, os.cpu_count() or 8))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dmosopt_amd import _hipops as ext
dev = torch.device("cuda")


def bench(fn, reps=9, inner=50):
    """Median of reps, each timing `inner` back-to-back launches (amortizes
    the ~quarter-millisecond host launch+sync overhead that otherwise
    swamps a 30-100 us kernel)."""
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(reps):
        t0 = time.perf_counter()
        for _ in range(inner):
            fn()
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) / inner)
    ts.sort()
    return 1e6 * ts[len(ts) // 2]


for (B, N, R) in [(64, 300, 1), (96, 300, 1), (2, 300, 200), (2, 300, 1024)]:
    A = torch.randn(B, N, N, device=dev) * 0.1
    K = A @ A.transpose(-1, -2) + 10.0 * torch.eye(N, device=dev)
    L = torch.linalg.cholesky(K.cpu()).to(dev).contiguous()
    Y0 = torch.randn(B, N, R, device=dev)

    Yw = Y0.clone()

    def fwd():
        ext.forward_solve_(L, Yw)

    def bwd():
        ext.backward_solve_(L, Yw)

    # correctness spot-check
    Z = Y0.clone()
    ext.forward_solve_(L, Z)
    ref = torch.linalg.solve_triangular(L, Y0, upper=False)
    err = (Z - ref).abs().max().item() / ref.abs().max().item()
    print(f"B={B} N={N} R={R}: fwd {bench(fwd):8.1f} us  bwd {bench(bwd):8.1f} us  relerr {err:.2e}")
