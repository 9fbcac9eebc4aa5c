"""Batched SCE-UA hyperparameter search.

Shuffled-Complex-Evolution (Duan 2004) with the reference's parameters and
control flow (reference model.py:1419-1753: npg=2n+1, nps=n+1, nspl=npg,
simplex selection via linear probability, reflect->contract->random CCE
step, geometric-range and criterion-change convergence) — re-designed for
GPU execution:

* S independent streams (one per objective) run in lockstep.
* Complexes are independent between shuffles, so ALL streams x ALL
  complexes advance together: each CCE stage is ONE batched call
  func((S*G, nopt), stream_ids) -> (S*G,) — on GPU a single batched
  kernel-build + batched Cholesky launch instead of thousands of
  sequential N x N factorizations.
* The reflect->contract->random cascade is mask-batched.

Evaluation ORDER differs from the sequential reference (statistical parity,
not bitwise — the accepted RNG discipline per SURVEY.md section 7).
"""

from __future__ import annotations

from typing import Callable, Optional

import numpy as np
import torch


def _select_simplex(nps: int, npg: int, rng: np.random.Generator) -> np.ndarray:
    """Sample nps distinct positions in [0, npg) with linear (triangular)
    bias toward better-ranked points; position 0 always included
    (reference model.py:1503-1520)."""
    lcs = {0}
    for _ in range(1, nps):
        for _attempt in range(1000):
            u = rng.uniform()
            lpos = int(
                np.floor(npg + 0.5 - np.sqrt((npg + 0.5) ** 2 - npg * (npg + 1) * u))
            )
            lpos = min(max(lpos, 0), npg - 1)
            if lpos not in lcs:
                lcs.add(lpos)
                break
        else:
            for cand in range(npg):
                if cand not in lcs:
                    lcs.add(cand)
                    break
    return np.asarray(sorted(lcs), dtype=np.int64)


def sceua_batched(
    func: Callable[[torch.Tensor, torch.Tensor], torch.Tensor],
    bl: np.ndarray,
    bu: np.ndarray,
    nopt: int,
    n_streams: int = 1,
    ngs: Optional[int] = None,
    maxn: int = 3000,
    kstop: int = 10,
    pcento: float = 0.1,
    peps: float = 0.001,
    seed=None,
    logger=None,
    device=None,
    dtype=torch.float64,
):
    """Run S parallel SCE-UA minimizations with batched evaluation.

    func(x (B, nopt), stream (B,) long) -> (B,) objective values.
    Returns (bestx (S, nopt), bestf (S,), icall (S,)) as numpy arrays.
    """
    rng = np.random.default_rng(seed)
    device = device or torch.device("cpu")
    if ngs is None:
        ngs = nopt
    npg = 2 * nopt + 1
    nps = nopt + 1
    nspl = npg
    npt = npg * ngs
    S, G = n_streams, ngs
    bl_t = torch.as_tensor(bl, dtype=dtype, device=device)
    bu_t = torch.as_tensor(bu, dtype=dtype, device=device)
    bd = bu_t - bl_t

    # device-resident uniform draws (seeded once from the host Generator):
    # avoids a host->device copy per CCE stage
    tgen = torch.Generator(device=device)
    tgen.manual_seed(int(rng.integers(0, 2**63 - 1)))

    def rand_points(shape):
        u = torch.rand(shape + (nopt,), dtype=dtype, device=device, generator=tgen)
        return u * bd + bl_t

    x = rand_points((S, npt))  # (S, npt, nopt)
    ids_all = torch.arange(S, device=device).repeat_interleave(npt)
    xf = func(x.reshape(S * npt, nopt), ids_all).reshape(S, npt)
    icall = np.full(S, npt, dtype=np.int64)

    def sort_pop(x_, xf_):
        o = torch.argsort(xf_, dim=-1)
        return (
            torch.gather(x_, -2, o[..., None].expand(*o.shape, nopt)),
            torch.gather(xf_, -1, o),
        )

    x, xf = sort_pop(x, xf)

    # fused CCE stage path: propose + accept/resort run as two extension
    # kernels (Philox RNG in-kernel) instead of ~30 torch launches/stage
    use_native_stage = False
    if device.type == "cuda" and dtype == torch.float32:
        from dmosopt_amd import ops as _ops

        if _ops.native_available():
            lds = (npg * 2 + npg * nopt) * 4
            use_native_stage = lds <= 60 * 1024
    if use_native_stage:
        from dmosopt_amd import _hipops as _hip
        bl_f = bl_t.to(torch.float32).contiguous()
        bu_f = bu_t.to(torch.float32).contiguous()

    def gnrng_of(x_):
        rngs = (x_.max(dim=1).values - x_.min(dim=1).values) / bd
        return torch.exp(torch.log(rngs.clamp_min(1e-300)).mean(dim=1))

    gnrng = gnrng_of(x).cpu().numpy()
    criter = [[] for _ in range(S)]
    criter_change = np.full(S, 1e5)
    sid_sg = torch.arange(S, device=device).repeat_interleave(G)
    sid3 = torch.cat([sid_sg, sid_sg, sid_sg], dim=0)

    nloop = 0
    while True:
        act_np = (icall < maxn) & (gnrng > peps) & (criter_change > pcento)
        if not act_np.any():
            break
        nloop += 1
        act = torch.as_tensor(act_np, device=device)

        # all simplex selections of this shuffle drawn up front: ONE small
        # H2D instead of one per CCE stage (a pageable-memory copy inside
        # the stage loop near-synchronizes the stream and kills pipelining)
        lcs_all = torch.as_tensor(
            np.stack([_select_simplex(nps, npg, rng) for _ in range(nspl)]).astype(
                np.int32
            ),
            device=device,
        )

        # partition: position p of complex g sits at row p*ngs+g
        cx = x.view(S, npg, G, nopt).permute(0, 2, 1, 3).contiguous()  # (S,G,npg,nopt)
        cf = xf.view(S, npg, G).permute(0, 2, 1).contiguous()  # (S,G,npg)
        icall_dev = torch.zeros(S, dtype=torch.float64, device=device)

        if use_native_stage:
            act_i32 = act.to(torch.int32).contiguous()
            icall_i32 = torch.zeros(S, dtype=torch.int32, device=device)
            for _step in range(nspl):
                lcs = lcs_all[_step].contiguous()
                cand = _hip.sceua_propose(
                    cx, lcs, bl_f, bu_f, nps, int(rng.integers(0, 2**62))
                )
                fall = func(cand.reshape(3 * S * G, nopt), sid3)
                ok = _hip.sceua_accept(
                    cx, cf, cand, fall.to(torch.float32).contiguous(), lcs,
                    act_i32, icall_i32, nps,
                )
                assert ok
            icall_dev = icall_i32.to(torch.float64)
        else:
          for _step in range(nspl):
            lcs = lcs_all[_step].long()
            s_pts = cx[:, :, lcs, :]  # (S,G,nps,nopt)
            s_f = cf[:, :, lcs]  # (S,G,nps)

            sw = s_pts[:, :, -1, :]  # (S,G,nopt) worst
            fw = s_f[:, :, -1]  # (S,G)
            ce = s_pts[:, :, :-1, :].mean(dim=2)

            # SPECULATIVE CCE: evaluate reflection, contraction and random
            # fallback for every complex in ONE batched call, select on
            # device with the sequential acceptance rule (reflect unless it
            # fails the worst; then contraction; then random). No host syncs
            # inside the stage; the actual-evaluation count is accumulated
            # on device with the sequential semantics.
            s_ref = ce + (ce - sw)
            oob = ((s_ref < bl_t) | (s_ref > bu_t)).any(dim=-1, keepdim=True)
            s_ref = torch.where(oob, rand_points((S, G)), s_ref)
            s_con = sw + 0.5 * (ce - sw)
            s_rnd = rand_points((S, G))
            cand = torch.cat(
                [s_ref.reshape(S * G, nopt), s_con.reshape(S * G, nopt),
                 s_rnd.reshape(S * G, nopt)], dim=0)
            fall = func(cand, sid3).reshape(3, S, G)
            f_ref, f_con, f_rnd = fall[0], fall[1], fall[2]

            use_con = f_ref > fw
            use_rnd = use_con & (f_con > fw)
            fnew = torch.where(use_rnd, f_rnd, torch.where(use_con, f_con, f_ref))
            snew = torch.where(
                use_rnd[..., None], s_rnd,
                torch.where(use_con[..., None], s_con, s_ref))
            icall_dev += (
                G + use_con.sum(dim=1).to(torch.float64)
                + use_rnd.sum(dim=1).to(torch.float64)
            )

            # replace worst simplex point (only for active streams), reinsert
            upd = act[:, None]
            s_pts = s_pts.clone()
            s_f = s_f.clone()
            s_pts[:, :, -1, :] = torch.where(upd[..., None], snew, s_pts[:, :, -1, :])
            s_f[:, :, -1] = torch.where(upd, fnew, s_f[:, :, -1])
            cx[:, :, lcs, :] = s_pts
            cf[:, :, lcs] = s_f
            cx, cf = sort_pop(cx, cf)

        x = cx.permute(0, 2, 1, 3).reshape(S, npt, nopt)
        xf = cf.permute(0, 2, 1).reshape(S, npt)
        x, xf = sort_pop(x, xf)
        # ONE device->host transfer per shuffle for every termination input
        pack = torch.cat(
            [icall_dev, gnrng_of(x), xf[:, 0].to(torch.float64)]
        ).cpu().numpy()
        icall += pack[:S].astype(np.int64) * act_np
        gnrng = pack[S : 2 * S]
        bestf_now = pack[2 * S :]
        for s in range(S):
            if act_np[s]:
                criter[s].append(float(bestf_now[s]))
                nl = len(criter[s])
                if nl >= kstop:
                    num = abs(criter[s][-1] - criter[s][-kstop]) * 100.0
                    den = np.mean(np.abs(criter[s][-kstop:]))
                    criter_change[s] = num / den if den > 0 else 0.0
        if logger is not None:
            logger.info(
                f"sceua_batched loop {nloop}: bestf={bestf_now}, icall={icall}"
            )

    return x[:, 0, :].cpu().numpy(), xf[:, 0].cpu().numpy(), icall
