"""Joint FT-Transformer surrogate (reference model_transformer.py:161-1233).

PyTorch-native (the reference uses Keras 3): per-parameter feature
tokenization + CLS token, pre-LN transformer blocks, joint heads for
objective regression and constraint-feasibility classification, bounds
input normalization, range/zscore target normalization, holdout-validated
epoch selection ('auto'), and autograd-based sensitivity. The ``joint``
function is the ``surrogate_custom_training`` hook with the reference's
return contract: (optimizer_cls, objective_model, feasibility_model,
sensitivity_provider).
"""

from __future__ import annotations

import logging
import math
from typing import Dict, Optional

import numpy as np
import torch
import torch.nn as nn

logger = logging.getLogger("dmosopt_amd.transformer")


class TransformerBlock(nn.Module):
    def __init__(self, dim, ff_dim, num_heads, attention_dropout=0.1,
                 ffn_dropout=0.05, residual_dropout=0.0, input_norm=True):
        super().__init__()
        self.att_norm = nn.LayerNorm(dim) if input_norm else None
        self.mha = nn.MultiheadAttention(
            dim, num_heads, dropout=attention_dropout, batch_first=True
        )
        self.att_drop = nn.Dropout(residual_dropout)
        self.ffn_norm = nn.LayerNorm(dim)
        self.ffn = nn.Sequential(
            nn.Linear(dim, ff_dim), nn.GELU(), nn.Dropout(ffn_dropout), nn.Linear(ff_dim, dim)
        )
        self.ffn_drop = nn.Dropout(residual_dropout)

    def forward(self, x):
        h = self.att_norm(x) if self.att_norm is not None else x
        a, _ = self.mha(h, h, h, need_weights=False)
        x = x + self.att_drop(a)
        x = x + self.ffn_drop(self.ffn(self.ffn_norm(x)))
        return x


class JointFTTransformer(nn.Module):
    """FT-Transformer over parameter feature tokens with joint objective +
    constraint heads."""

    def __init__(
        self,
        num_parameters: int,
        num_constraints: int,
        num_objectives: int,
        mode: str = "c+o",
        xlb=None,
        xub=None,
        learning_rate: float = 1e-3,
        normalize_targets: str = "range",
        n_blocks: int = 3,
        embedding_dim_per_head: int = 32,
        num_heads: int = 4,
        ffn_ratio: float = 2.0,
        device=None,
        seed: Optional[int] = None,
        **kwargs,
    ):
        super().__init__()
        if mode not in ("c+o", "c", "o"):
            raise ValueError("Invalid mode")
        self.num_parameters = num_parameters
        self.num_constraints = num_constraints
        self.num_objectives = num_objectives
        self.mode = mode
        self.learning_rate = learning_rate
        self.normalize_targets = normalize_targets
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        if seed is not None:
            torch.manual_seed(seed)

        dim = embedding_dim_per_head * num_heads
        self.dim = dim
        if xlb is not None and xub is not None:
            lb = torch.as_tensor(np.asarray(xlb, dtype=np.float32))
            rg = torch.as_tensor(np.asarray(xub, dtype=np.float32)) - lb
            rg[rg == 0] = 1.0
        else:
            lb = torch.zeros(num_parameters)
            rg = torch.ones(num_parameters)
        self.register_buffer("xlb_t", lb)
        self.register_buffer("xrg_t", rg)

        # per-parameter scalar tokenization: token_i = x_i * W_i + b_i
        self.tok_w = nn.Parameter(torch.randn(num_parameters, dim) * 0.02)
        self.tok_b = nn.Parameter(torch.zeros(num_parameters, dim))
        self.cls = nn.Parameter(torch.randn(1, 1, dim) * 0.02)
        self.blocks = nn.ModuleList(
            [
                TransformerBlock(dim, int(dim * ffn_ratio), num_heads)
                for _ in range(n_blocks)
            ]
        )
        self.final_norm = nn.LayerNorm(dim)
        self.obj_head = nn.Linear(dim, num_objectives) if mode in ("c+o", "o") else None
        self.con_head = nn.Linear(dim, num_constraints) if (
            mode in ("c+o", "c") and num_constraints > 0
        ) else None
        self.register_buffer("y_shift", torch.zeros(num_objectives))
        self.register_buffer("y_scale", torch.ones(num_objectives))
        self._last_fit_epochs = -1
        self.stats: Dict = {}
        self.to(self.device)

    # ------------------------------------------------------------- forward
    def forward(self, x):
        xn = (x - self.xlb_t) / self.xrg_t
        tokens = xn[:, :, None] * self.tok_w[None, :, :] + self.tok_b[None, :, :]
        tokens = torch.cat([self.cls.expand(x.shape[0], 1, self.dim), tokens], dim=1)
        for blk in self.blocks:
            tokens = blk(tokens)
        h = self.final_norm(tokens[:, 0])
        out = {}
        if self.obj_head is not None:
            out["objectives"] = self.obj_head(h)
        if self.con_head is not None:
            out["constraints"] = torch.sigmoid(self.con_head(h))
        return out

    # ------------------------------------------------------------ training
    def _norm_y(self, y: torch.Tensor, adapt: bool):
        if adapt:
            if self.normalize_targets == "range":
                lo, hi = y.min(dim=0).values, y.max(dim=0).values
                self.y_shift.copy_(lo)
                self.y_scale.copy_((hi - lo).clamp_min(1e-12))
            else:  # zscore
                self.y_shift.copy_(y.mean(dim=0))
                self.y_scale.copy_(y.std(dim=0).clamp_min(1e-12))
        return (y - self.y_shift) / self.y_scale

    def objective_loss(self, pred, y_true_n, yC_true=None, alpha=1.0, beta=0.01):
        """Joint objective MSE + constraint BCE (reference :424-449)."""
        loss = torch.zeros((), device=self.device)
        if "objectives" in pred and y_true_n is not None:
            loss = loss + alpha * nn.functional.mse_loss(pred["objectives"], y_true_n)
        if "constraints" in pred and yC_true is not None:
            loss = loss + beta * nn.functional.binary_cross_entropy(
                pred["constraints"].clamp(1e-7, 1 - 1e-7), yC_true
            )
        return loss

    def fit(self, x, y, yC=None, epochs=300, batch_size=64, verbose=0):
        X = torch.as_tensor(np.asarray(x, dtype=np.float32), device=self.device)
        Yn = self._norm_y(
            torch.as_tensor(np.asarray(y, dtype=np.float32), device=self.device), adapt=True
        )
        YC = (
            torch.as_tensor(np.asarray(yC, dtype=np.float32), device=self.device)
            if yC is not None
            else None
        )
        opt = torch.optim.AdamW(self.parameters(), lr=self.learning_rate)
        n = X.shape[0]
        self.train()
        for ep in range(epochs):
            perm = torch.randperm(n, device=self.device)
            total = 0.0
            for s in range(0, n, batch_size):
                idx = perm[s : s + batch_size]
                opt.zero_grad(set_to_none=True)
                pred = self(X[idx])
                loss = self.objective_loss(
                    pred, Yn[idx], YC[idx] if YC is not None else None
                )
                loss.backward()
                opt.step()
                total += loss.item() * len(idx)
            if verbose and (ep % 50 == 0):
                logger.info(f"transformer epoch {ep}: loss {total / n:.5f}")
        self.eval()
        self._last_fit_epochs = epochs
        return self

    def autofit(self, x, y, yC=None, epochs="auto", verbose=0, **kwargs):
        """Fit with holdout-selected epoch count when epochs='auto'
        (reference autofit :500-577 uses CV-chosen epochs)."""
        if epochs != "auto":
            return self.fit(x, y, yC, epochs=int(epochs), verbose=verbose)
        x = np.asarray(x, dtype=np.float32)
        y = np.asarray(y, dtype=np.float32)
        n = x.shape[0]
        rng = np.random.default_rng(0)
        idx = rng.permutation(n)
        n_val = max(1, n // 5)
        val, tr = idx[:n_val], idx[n_val:]
        X_tr = torch.as_tensor(x[tr], device=self.device)
        Yn_tr = self._norm_y(torch.as_tensor(y[tr], device=self.device), adapt=True)
        X_val = torch.as_tensor(x[val], device=self.device)
        Yn_val = self._norm_y(torch.as_tensor(y[val], device=self.device), adapt=False)
        YC_tr = YC_val = None
        if yC is not None:
            yC = np.asarray(yC, dtype=np.float32)
            YC_tr = torch.as_tensor(yC[tr], device=self.device)
            YC_val = torch.as_tensor(yC[val], device=self.device)

        opt = torch.optim.AdamW(self.parameters(), lr=self.learning_rate)
        best_val, best_ep, patience = float("inf"), 50, 0
        max_epochs = 600
        self.train()
        for ep in range(1, max_epochs + 1):
            opt.zero_grad(set_to_none=True)
            loss = self.objective_loss(self(X_tr), Yn_tr, YC_tr)
            loss.backward()
            opt.step()
            if ep % 10 == 0:
                with torch.no_grad():
                    self.eval()
                    vl = float(self.objective_loss(self(X_val), Yn_val, YC_val))
                    self.train()
                if vl < best_val - 1e-6:
                    best_val, best_ep, patience = vl, ep, 0
                else:
                    patience += 1
                    if patience >= 10:
                        break
        # refit on the full data for the selected number of epochs
        for p in self.parameters():
            if p.dim() > 1:
                nn.init.xavier_uniform_(p)
        with torch.no_grad():
            self.cls.normal_(0, 0.02)
            self.tok_w.normal_(0, 0.02)
            self.tok_b.zero_()
        return self.fit(x, y, yC, epochs=best_ep, verbose=verbose)

    # ------------------------------------------------------------- predict
    @torch.no_grad()
    def predict(self, x, verbose=0):
        self.eval()
        X = torch.as_tensor(np.asarray(x, dtype=np.float32), device=self.device)
        out = self(X)
        res = {}
        if "objectives" in out:
            res["objectives"] = (
                out["objectives"] * self.y_scale + self.y_shift
            ).cpu().numpy()
        if "constraints" in out:
            res["constraints"] = out["constraints"].cpu().numpy()
        if self.mode == "c" and "constraints" in res:
            return res["constraints"]
        return res

    def predict_objectives(self, x, **kwargs):
        return self.predict(x)["objectives"]

    def evaluate(self, x):
        return self.predict_objectives(x)

    def autoeval(self, x, y, yC=None, verbose=0):
        pred = self.predict(x)
        scores = {}
        if "objectives" in pred:
            scores["objective_mae"] = np.mean(np.abs(pred["objectives"] - np.asarray(y)))
        if "constraints" in pred and yC is not None:
            acc = np.mean((pred["constraints"] > 0.5) == (np.asarray(yC) > 0.5))
            scores["constraint_accuracy"] = acc
        return scores

    # --------------------------------------------------------- sensitivity
    def sensitivity(self, points, reduction=None, key="objectives"):
        """Input-gradient sensitivity (reference :1006-1081): mean-squared
        d output / d input over the sample set."""
        X = torch.as_tensor(np.asarray(points, dtype=np.float32), device=self.device)
        X.requires_grad_(True)
        out = self(X)[key]
        grads = []
        for j in range(out.shape[1]):
            g = torch.autograd.grad(out[:, j].sum(), X, retain_graph=j < out.shape[1] - 1)[0]
            grads.append(g)
        G = torch.stack(grads, dim=0)  # (m, N, d)
        if reduction is None:
            sens = (G**2).mean(dim=1).max(dim=0).values
        else:
            sens = torch.as_tensor(
                np.max([np.asarray(reduction(g.detach().cpu().numpy())) for g in G], axis=0)
            )
        return {key: sens.detach().cpu().numpy()}


def joint(
    optimizer_cls, Xinit, Yinit, C, xlb, xub, file_path, options,
    mode="c+o", objectives=True, constraints=False, sensitivity=True,
    epochs="auto", iterations=[],
):
    """surrogate_custom_training hook (reference model_transformer.py:1112)."""
    x = np.asarray(Xinit).copy()
    y = np.asarray(Yinit).copy()
    yC = (np.asarray(C) > 0).astype(int) if C is not None else None

    class _Model:
        def __init__(self, model):
            self._wrapped = model
            self.stats = {}

        def rank(self, xq):
            if self._wrapped.num_constraints == 0:
                return np.ones(np.asarray(xq).shape[0])
            result = self._wrapped.predict(xq)
            probs = np.asarray(
                result["constraints"] if isinstance(result, dict) else result
            )
            return np.mean(probs, axis=1)

        def evaluate(self, xq):
            return self._wrapped.predict_objectives(xq)

        def predict(self, xq):
            mean = self._wrapped.predict_objectives(xq)
            return mean, np.zeros_like(mean)

        def di_dict(self):
            rng = np.random.default_rng(1)
            pts = rng.random((2048, len(xlb))) * (np.asarray(xub) - np.asarray(xlb)) + np.asarray(xlb)
            sens = self._wrapped.sensitivity(pts)["objectives"]
            sens = sens / (np.max(sens) + 1e-7)
            di = np.clip(1 + np.abs(sens) * 20, 1, 30).astype(np.float64)
            return {"di_mutation": di, "di_crossover": di}

        def __getattr__(self, name):
            return getattr(self._wrapped, name)

    model = _Model(
        JointFTTransformer(
            num_parameters=x.shape[1],
            num_constraints=C.shape[1] if C is not None else 0,
            num_objectives=y.shape[1],
            mode=mode,
            xlb=xlb,
            xub=xub,
        )
    )
    model._wrapped.autofit(x, y, yC, epochs=epochs)
    scores = model._wrapped.autoeval(x, y, yC)
    scores["num_samples"] = x.shape[0]
    scores["iteration"] = len(iterations)
    model.stats = {f"model_{k}": float(np.mean(v)) for k, v in scores.items()}
    return (
        optimizer_cls,
        model if objectives else None,
        model if constraints else None,
        model if sensitivity else None,
    )
