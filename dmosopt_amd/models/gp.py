"""GP surrogate models (surrogate duck-type: __init__(xin, yin, nInput,
nOutput, xlb, xub, **kw), predict(x) -> (mean, var), evaluate(x)).

GPRMatern ('gpr', the default) mirrors the reference GPR_Matern
(model.py:1182-1275): per-objective exact GP with ConstantKernel x
Matern(nu=2.5) + WhiteKernel, inputs normalized to [0,1], normalize_y,
hyperparameters from SCE-UA over the negative log marginal likelihood — but
all objectives fit as ONE batched SCE-UA run (models/sceua.py) and all
posterior math is batched over objectives (models/gp_core.py), targeting a
handful of large device launches per epoch.

EGPMatern ('egp') is the gradient-based alternative (reference
model_gpytorch.py:1929-2235's role): Adam on the exact MLL via autograd,
ARD lengthscales by default.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from dmosopt_amd import ops
from dmosopt_amd.models.gp_core import FittedGP, batched_nmll
from dmosopt_amd.models.sceua import sceua_batched


def _top_k_mo(x: np.ndarray, y: np.ndarray, top_k):
    """Keep the top_k rows by non-dominated sort (reference MOEA.py:350-372)."""
    xt, yt = ops.top_k_mo(
        torch.as_tensor(x, dtype=torch.float64),
        torch.as_tensor(y, dtype=torch.float64),
        top_k,
    )
    return xt.numpy() if xt is not x else x, yt.numpy() if yt is not y else y


class _GPRBase:
    nu: float = 2.5
    #: fitted state is fully determined by (archive, theta): the engine may
    #: run the hyperparameter search on rank 0 only and broadcast theta
    #: (core/engine.py train())
    supports_theta_broadcast = True

    def __init__(
        self,
        xin,
        yin,
        nInput,
        nOutput,
        xlb,
        xub,
        optimizer="sceua",
        seed=None,
        length_scale_bounds=(1e-3, 100.0),
        constant_kernel_bounds=(1e-4, 1e3),
        noise_level_bounds=(1e-9, 1e-2),
        anisotropic=False,
        return_mean_variance=False,
        nan="remove",
        top_k=None,
        adam_lr=0.08,
        adam_iters=150,
        batch_size=None,
        device=None,
        dtype=None,
        compute="fp32",
        logger=None,
        theta_override=None,
        **kwargs,
    ):
        self.nInput = nInput
        self.nOutput = nOutput
        self.xlb = np.asarray(xlb, dtype=np.float64)
        self.xub = np.asarray(xub, dtype=np.float64)
        self.xrg = self.xub - self.xlb
        self.xrg[self.xrg == 0] = 1.0
        self.logger = logger
        self.return_mean_variance = return_mean_variance
        self.anisotropic = anisotropic
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        self.dtype = dtype or (torch.float64 if self.device.type == "cpu" else torch.float32)
        # "bf16": bf16-MFMA posterior path (predictions + trailing Cholesky
        # updates); the hyperparameter SEARCH below is always fp32
        self.compute = compute

        xin = np.asarray(xin, dtype=np.float64)
        yin = np.asarray(yin, dtype=np.float64)
        if yin.ndim == 1:
            yin = yin.reshape(-1, 1)
        if nan is not None:
            yt, xt = ops.filter_samples(
                torch.as_tensor(yin), torch.as_tensor(xin), nan=nan
            )
            xin, yin = xt.numpy(), yt.numpy()
        xin, yin = _top_k_mo(xin, yin, top_k)
        yin = np.nan_to_num(yin)

        xn = (xin - self.xlb[None, :]) / self.xrg[None, :]
        X = torch.as_tensor(xn, dtype=self.dtype, device=self.device)
        Y = torch.as_tensor(yin, dtype=self.dtype, device=self.device)
        y_mean = Y.mean(dim=0)
        y_std = Y.std(dim=0, unbiased=False).clamp_min(1e-12)

        m = nOutput
        n_ell = nInput if anisotropic else 1
        nopt = 1 + n_ell + 1
        bl = np.concatenate(
            [
                [math.log(constant_kernel_bounds[0])],
                [math.log(length_scale_bounds[0])] * n_ell,
                [math.log(noise_level_bounds[0])],
            ]
        )
        bu = np.concatenate(
            [
                [math.log(constant_kernel_bounds[1])],
                [math.log(length_scale_bounds[1])] * n_ell,
                [math.log(noise_level_bounds[1])],
            ]
        )
        Yn = (Y - y_mean[None, :]) / y_std[None, :]

        if theta_override is not None:
            # broadcast-received hyperparameters: skip the search, rebuild
            # the posterior (Cholesky + alpha) locally from the archive
            theta = torch.as_tensor(
                np.asarray(theta_override), dtype=self.dtype, device=self.device
            )
            assert theta.shape == (m, nopt), (theta.shape, (m, nopt))
            self._fitted = FittedGP(
                X, Y, theta, y_mean, y_std, nu=self.nu, anisotropic=anisotropic,
                jitter=1e-10 if self.dtype == torch.float64 else 1e-6,
                compute=self.compute,
            )
            self.theta = theta
            return

        if optimizer == "dlib" and logger is not None:
            logger.warning(
                "GP optimizer 'dlib' (reference model.py:1367-1416 optional "
                "dlib.find_min_global path) is served by the batched SCE-UA "
                "search here; dlib is not a dependency of this framework"
            )
        if optimizer in ("sceua", "dlib", None):
            YnT = Yn.T.contiguous()  # (m, N)

            def nmll_func(theta_batch: torch.Tensor, stream: torch.Tensor):
                # one fused batched call for ALL streams: per-row y targets
                return batched_nmll(
                    X, YnT[stream], theta_batch.to(self.dtype),
                    nu=self.nu, anisotropic=anisotropic,
                )

            bestx, bestf, icall = sceua_batched(
                nmll_func, bl, bu, nopt, n_streams=m, seed=seed,
                device=self.device, dtype=self.dtype, logger=None,
            )
            theta = torch.as_tensor(bestx, dtype=self.dtype, device=self.device)
        elif optimizer == "adam":
            theta = self._fit_adam(X, Yn, bl, bu, nopt, m, adam_lr, adam_iters, seed)
        else:
            raise ValueError(f"Unknown GP optimizer {optimizer!r}")

        self._fitted = FittedGP(
            X, Y, theta, y_mean, y_std, nu=self.nu, anisotropic=anisotropic,
            jitter=1e-10 if self.dtype == torch.float64 else 1e-6,
            compute=self.compute,
        )
        self.theta = theta

    def _fit_adam(self, X, Yn, bl, bu, nopt, m, lr, iters, seed):
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(int(seed))
        bl_t = torch.as_tensor(bl, dtype=self.dtype, device=self.device)
        bu_t = torch.as_tensor(bu, dtype=self.dtype, device=self.device)
        init = 0.5 * (bl_t + bu_t)[None, :].repeat(m, 1)
        # sensible inits: sf2=1, ell=0.5, noise=1e-6 (reference defaults)
        init[:, 0] = 0.0
        init[:, 1:-1] = math.log(0.5)
        init[:, -1] = math.log(1e-6)
        theta = init.clone().requires_grad_(True)
        opt = torch.optim.Adam([theta], lr=lr)
        best_theta = init.clone()
        best_loss = torch.full((m,), float("inf"), dtype=self.dtype, device=self.device)
        for _ in range(iters):
            opt.zero_grad(set_to_none=True)
            losses = []
            for s in range(m):
                nm = batched_nmll(
                    X, Yn[:, s], theta[s : s + 1], nu=self.nu,
                    anisotropic=self.anisotropic, jitter=1e-8,
                    differentiable=True,
                )
                losses.append(nm)
            loss_vec = torch.cat(losses)
            finite = torch.isfinite(loss_vec)
            if not bool(finite.any()):
                break
            with torch.no_grad():
                better = finite & (loss_vec < best_loss)
                best_loss = torch.where(better, loss_vec.detach(), best_loss)
                best_theta[better] = theta.detach()[better]
            loss_vec.masked_fill(~finite, 0.0).sum().backward()
            if not torch.isfinite(theta.grad).all():
                theta.grad.zero_()
            opt.step()
            with torch.no_grad():
                theta.clamp_(bl_t, bu_t)
                theta.nan_to_num_(nan=0.0)
        return best_theta

    # ---------------------------------------------------------------- API
    def predict(self, xin):
        xin = np.asarray(xin, dtype=np.float64)
        if xin.ndim == 1:
            xin = xin.reshape(1, self.nInput)
        xn = (xin - self.xlb[None, :]) / self.xrg[None, :]
        Xq = torch.as_tensor(xn, dtype=self.dtype, device=self.device)
        mean, var = self._fitted.predict(Xq)
        return mean.cpu().numpy().astype(np.float64), var.cpu().numpy().astype(np.float64)

    def predict_tensor(self, Xq_normalized: torch.Tensor):
        """Device-resident predict for the inner MOEA loop: takes ALREADY
        normalized inputs on the model device, returns device tensors."""
        return self._fitted.predict(Xq_normalized)

    def normalize_query(self, xin: torch.Tensor) -> torch.Tensor:
        # cache the bounds tensors: re-uploading two tiny arrays is two H2D
        # transfers per surrogate evaluation (every generation)
        cache = getattr(self, "_nq_cache", None)
        key = (xin.dtype, xin.device)
        if cache is None or cache[0] != key:
            lb = torch.as_tensor(self.xlb, dtype=xin.dtype, device=xin.device)
            rg = torch.as_tensor(self.xrg, dtype=xin.dtype, device=xin.device)
            self._nq_cache = cache = (key, lb, rg)
        return (xin - cache[1]) / cache[2]

    def evaluate(self, x):
        mean, var = self.predict(x)
        if self.return_mean_variance:
            return mean, var
        return mean

    def evaluate_tensor(self, x: torch.Tensor) -> torch.Tensor:
        """Device-resident evaluate: tensor in, tensor out, no host trip."""
        xr = x.to(self.device, self.dtype)
        if (
            not self.return_mean_variance
            and xr.device.type == "cuda"
            and self.compute != "bf16"
        ):
            from dmosopt_amd import ops

            if ops.native_available():
                # raw queries: the per-dim normalization happens inside the
                # cross-kernel load (two fewer launches per generation)
                cache = getattr(self, "_affine_cache", None)
                if cache is None or cache[0] != xr.device:
                    lb = torch.as_tensor(self.xlb, dtype=torch.float32, device=xr.device)
                    invrg = torch.as_tensor(
                        1.0 / self.xrg, dtype=torch.float32, device=xr.device
                    )
                    self._affine_cache = cache = (xr.device, lb, invrg)
                from dmosopt_amd import _hipops

                f = self._fitted
                # static per-fit args cached once: the generation loop is
                # host-dispatch-bound, and re-checking contiguity/dtype of
                # six unchanged tensors every call costs ~8 us of host time
                pa = getattr(f, "_pred_args", None)
                if pa is None:
                    pa = f._pred_args = (
                        f.X.contiguous(), f.theta.contiguous().float(),
                        f.alpha.contiguous().float(), f.y_mean.float(),
                        f.y_std.float(),
                        0.0 if (f.nu is None or f.nu == float("inf")) else float(f.nu),
                        bool(f.anisotropic),
                    )
                return _hipops.gp_predict_mean(
                    xr.float().contiguous(), pa[0], pa[1], pa[2], pa[3],
                    pa[4], pa[5], pa[6], cache[1], cache[2],
                )
        xq = self.normalize_query(xr)
        mean, _ = self._fitted.predict(xq, return_var=self.return_mean_variance)
        return mean


class GPRMatern(_GPRBase):
    """Registry name 'gpr' — Matern-5/2 exact GP (reference model.py:1182)."""

    nu = 2.5


class GPRRBF(_GPRBase):
    """RBF kernel variant (reference model.py:1278 GPR_RBF)."""

    nu = float("inf")


class EGPMatern(_GPRBase):
    """Registry name 'egp' — gradient-fit exact GP, ARD lengthscales.

    Plays the role of the reference's GPyTorch EGP_Matern
    (model_gpytorch.py:1929-2235): Adam on the exact marginal likelihood.
    """

    nu = 2.5

    def __init__(self, xin, yin, nInput, nOutput, xlb, xub, **kwargs):
        kwargs.setdefault("optimizer", "adam")
        kwargs.setdefault("anisotropic", True)
        super().__init__(xin, yin, nInput, nOutput, xlb, xub, **kwargs)


class MEGPMatern(_GPRBase):
    """Registry name 'megp' — multitask exact GP.

    Round-1 implementation: batched per-task exact GPs with shared ARD
    kernel hyperparameters fit jointly (sum of per-task MLLs), standing in
    for the reference's Kronecker multitask GP (model_gpytorch.py:1623-1928).
    Full task-covariance (ICM) model is planned.
    """

    nu = 2.5

    def __init__(self, xin, yin, nInput, nOutput, xlb, xub, **kwargs):
        kwargs.setdefault("optimizer", "adam")
        kwargs.setdefault("anisotropic", True)
        super().__init__(xin, yin, nInput, nOutput, xlb, xub, **kwargs)
