"""A/B: single-block vs multi-launch cholesky at SCE-UA batch shapes (same box)."""
import os, sys, time, subprocess

if len(sys.argv) > 1:
    import torch
    torch.set_num_threads(8)
    sys.path.insert(0, "/root/repo")
    from dmosopt_amd import _hipops as ext
    dev = torch.device("cuda")
    for B in (6, 12, 24, 48, 96, 192):
        N = 300
        A = torch.randn(B, N, N, device=dev) * 0.1
        K = (A @ A.transpose(-1, -2) + 10.0 * torch.eye(N, device=dev)).contiguous()
        # correctness
        W = K.clone(); ld, info = ext.cholesky_batched_(W)
        Lref = torch.linalg.cholesky(K.cpu().double())
        err = (torch.tril(W.cpu().double()) - Lref).abs().max().item()
        ld_err = (ld.cpu().double() - Lref.diagonal(dim1=1, dim2=2).log().sum(1)).abs().max().item()
        def run():
            W = K.clone()
            ext.cholesky_batched_(W)
        for _ in range(5): run()
        torch.cuda.synchronize()
        ts = []
        for _ in range(9):
            t0 = time.perf_counter()
            for _ in range(20): run()
            torch.cuda.synchronize()
            ts.append((time.perf_counter() - t0) / 20)
        ts.sort()
        print(f"mode={os.environ.get('DMOSOPT_CHOL_MODE','auto'):7s} B={B:4d}: {1e6*ts[4]:9.1f} us  err={err:.2e} ld_err={ld_err:.2e} info={int(info.abs().sum())}")
else:
    for mode in ("single", "multik"):
        env = dict(os.environ, DMOSOPT_CHOL_MODE=mode)
        subprocess.run([sys.executable, __file__, "go"], env=env, check=False)
