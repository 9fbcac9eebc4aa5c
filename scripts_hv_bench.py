"""Micro-benchmark the exact-HV device kernels vs the host implementations.

Run on a GPU box:
  python scripts_hv_bench.py            # timing table
  rocprofv3 --kernel-trace --stats -d gpurun_out/hvprof -- python scripts_hv_bench.py
"""

import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")

from dmosopt_amd.hv import exact


def bench(fn, *args, reps=20, warmup=3):
    for _ in range(warmup):
        fn(*args)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        out = fn(*args)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3, out


def np_hv2d(pts, ref):
    pts = pts[np.all(pts < ref, axis=1)]
    if len(pts) == 0:
        return 0.0
    order = np.lexsort((pts[:, 1], pts[:, 0]))
    pts = pts[order]
    y = pts[:, 1]
    pm = np.empty(len(y))
    pm[0] = np.inf
    pm[1:] = np.minimum.accumulate(y)[:-1]
    pts = pts[y < pm]
    if len(pts) == 0:
        return 0.0
    x_next = np.empty(len(pts))
    x_next[:-1] = pts[1:, 0]
    x_next[-1] = ref[0]
    return float(np.dot(x_next - pts[:, 0], ref[1] - pts[:, 1]))


def main():
    rng = np.random.default_rng(0)
    print(f"{'case':36s} {'host ms':>9s} {'device ms':>10s} {'ratio':>7s}")

    for n in (200, 1024, 4096):
        pts = rng.random((n, 2)) * 1.2
        ref = np.array([1.1, 1.1])
        th, vh = bench(np_hv2d, pts, ref)
        if torch.cuda.is_available():
            ptd = torch.as_tensor(pts, device="cuda")
            td, vd = bench(exact.hv_2d_device, ptd, ref)
            assert abs(vh - vd) < 1e-9 * max(1, abs(vh)), (vh, vd)
        else:
            td, vd = float("nan"), vh
        print(f"hv2d n={n:<6d}                      {th:9.3f} {td:10.3f} {th/td:7.1f}x")

    for n in (200, 1024):
        pts = rng.random((n, 3)) * 1.2
        ref = np.array([1.1, 1.1, 1.1])

        def host3():
            pf = pts[np.all(pts < ref, axis=1)]
            pf = pf[np.argsort(pf[:, 2], kind="stable")]
            tot = 0.0
            for i in range(len(pf)):
                z_hi = pf[i + 1, 2] if i + 1 < len(pf) else ref[2]
                dz = z_hi - pf[i, 2]
                if dz > 0:
                    tot += dz * np_hv2d(pf[: i + 1, :2], ref[:2])
            return tot

        th, vh = bench(host3, reps=5)
        if torch.cuda.is_available():
            td, vd = bench(exact.hv_3d_device, pts, ref)
            assert abs(vh - vd) < 1e-9 * max(1, abs(vh))
        else:
            td = float("nan")
        print(f"hv3d n={n:<6d}                      {th:9.3f} {td:10.3f} {th/td:7.1f}x")

    for d, n in ((4, 128), (5, 128), (6, 96)):
        ref = np.full(d, 1.0 + 0.1 * d)
        pts = rng.random((n, d))
        p = exact._filter_dominated(pts)
        p = p[np.all(p < ref, axis=1)]
        p = p[np.argsort(p[:, -1], kind="stable")]

        def host_lacour():
            ub = exact._FlatUBSet(ref, p)
            for i in range(len(p)):
                ub.insert(i)
            return float(ub.volumes().sum())

        th, vh = bench(host_lacour, reps=3, warmup=1)
        if torch.cuda.is_available():
            td, vd = bench(exact.lacour_hv_device, p, ref, reps=3, warmup=1)
            assert abs(vh - vd) < 1e-9 * max(1, abs(vh)), (vh, vd)
        else:
            td = float("nan")
        print(f"lacour d={d} n={len(p):<5d}                  {th:9.3f} {td:10.3f} {th/td:7.1f}x")

    for B, nfront, d in ((512, 100, 2), (4096, 200, 2), (4096, 200, 5)):
        ref = np.full(d, 2.0)
        box = exact.HyperVolumeBoxDecomposition(ref)
        front = rng.random((nfront, d))
        L, U = box._decompose_dominated_space(front)
        mu = rng.random((B, d)) * 1.5
        var = rng.random((B, d)) * 0.2 + 0.01

        def host_ehvi():
            # force the host path: _batch_ehvi would route to the device
            saved = exact._device_ready
            exact._device_ready = lambda: False
            try:
                return box._batch_ehvi(L, U, mu, var)
            finally:
                exact._device_ready = saved

        th, vh = bench(host_ehvi, reps=5)
        if torch.cuda.is_available():
            td, vd = bench(box._batch_ehvi_device, L, U, mu, var, reps=5)
            np.testing.assert_allclose(vd, vh, rtol=1e-4, atol=1e-7)
        else:
            td = float("nan")
        print(f"ehvi B={B:<5d} boxes={L.shape[0]:<5d} d={d}       {th:9.3f} {td:10.3f} {th/td:7.1f}x")


if __name__ == "__main__":
    main()
