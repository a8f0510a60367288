"""PILCO models: exact-GP world model + RBF controller.

Reference: pytorch/rl torchrl/modules/models/gp.py:31 (GPWorldModel —
one GP per state dim predicting transition residuals, Deisenroth &
Rasmussen 2011) and RBFController.

rl_amd form: the reference requires gpytorch/botorch (absent in this
image); here the exact RBF-kernel GP is ~80 lines of pure torch
(Cholesky of K + sigma^2 I, closed-form posterior, marginal-likelihood
hyperparameter fit with Adam), so the PILCO vertical runs self-contained
on the MI355X stack.  Belief propagation uses the GP posterior at the
input mean with the predictive variance inflated by the input
uncertainty through the kernel's expected value (exact for the mean
term of the moment-matching equations; the cross-covariance term of
Eqs. 20-23 is omitted — documented deviation).
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
from torch import nn

__all__ = ["ExactGPRegressor", "GPWorldModel", "RBFController"]


class ExactGPRegressor(nn.Module):
    """Single-output exact GP with an ARD RBF kernel (pure torch)."""

    def __init__(self, in_dim: int):
        super().__init__()
        self.log_lengthscale = nn.Parameter(torch.zeros(in_dim))
        self.log_signal = nn.Parameter(torch.zeros(()))
        self.log_noise = nn.Parameter(torch.tensor(-2.0))
        self._X: Optional[torch.Tensor] = None
        self._alpha: Optional[torch.Tensor] = None
        self._L: Optional[torch.Tensor] = None

    def _kernel(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        ls = self.log_lengthscale.exp()
        d = (a.unsqueeze(-2) / ls - b.unsqueeze(-3) / ls).pow(2).sum(-1)
        return (2 * self.log_signal).exp() * torch.exp(-0.5 * d)

    def _gram(self, X: torch.Tensor) -> torch.Tensor:
        n = X.shape[0]
        noise = (2 * self.log_noise).exp().clamp_min(1e-6)
        # jitter scaled by the signal variance: keeps the Cholesky
        # well-conditioned when the optimizer drives lengthscales large
        jitter = 1e-6 * (2 * self.log_signal).exp().detach() + 1e-8
        return self._kernel(X, X) + (noise + jitter) * torch.eye(n, device=X.device)

    def nll(self, X: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        n = X.shape[0]
        K = self._gram(X)
        L = torch.linalg.cholesky(K)
        alpha = torch.cholesky_solve(y.unsqueeze(-1), L).squeeze(-1)
        return (
            0.5 * (y * alpha).sum()
            + torch.log(torch.diagonal(L)).sum()
            + 0.5 * n * math.log(2 * math.pi)
        )

    def fit(self, X: torch.Tensor, y: torch.Tensor, iters: int = 50, lr: float = 0.1):
        # exact-GP linear algebra in float64: n is small (PILCO-scale),
        # and float32 Cholesky fails on near-singular RBF Grams
        self.double()
        X = X.double()
        y = y.double()
        opt = torch.optim.Adam(self.parameters(), lr=lr)
        for _ in range(iters):
            opt.zero_grad()
            loss = self.nll(X, y)
            loss.backward()
            opt.step()
        with torch.no_grad():
            K = self._gram(X)
            self._L = torch.linalg.cholesky(K)
            self._alpha = torch.cholesky_solve(y.unsqueeze(-1), self._L).squeeze(-1)
            self._X = X
        return self

    def predict(self, Xq: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        if self._X is None:
            raise RuntimeError("fit() the GP first")
        dtype = Xq.dtype
        k = self._kernel(Xq.double(), self._X)  # [Q, N]
        mean = k @ self._alpha
        v = torch.cholesky_solve(k.transpose(-2, -1), self._L)
        var = (2 * self.log_signal).exp() - (k * v.transpose(-2, -1)).sum(-1)
        return mean.to(dtype), var.clamp_min(1e-8).to(dtype)


class GPWorldModel(nn.Module):
    """One independent GP per state dim predicting the residual
    ``delta = x_{t+1} - x_t`` from ``[x, u]`` (PILCO Sec. 2.1).

    TensorDict interface matches the belief-space schema of
    :class:`~rl_amd.envs.transforms.MeanActionSelector`:
    reads ``("observation","mean"/"var")`` + ``("action","mean")``
    (falling back to flat ``observation``/``action``), writes
    ``("next","observation","mean"/"var")``.
    """

    def __init__(self, obs_dim: int, action_dim: int):
        super().__init__()
        self.obs_dim = obs_dim
        self.action_dim = action_dim
        self.gps = nn.ModuleList(
            [ExactGPRegressor(obs_dim + action_dim) for _ in range(obs_dim)]
        )

    def fit(self, dataset, iters: int = 50) -> "GPWorldModel":
        obs = dataset.get("observation")
        act = dataset.get("action")
        nxt = dataset.get(("next", "observation"))
        X = torch.cat([obs, act], dim=-1)
        delta = nxt - obs
        for d, gp in enumerate(self.gps):
            gp.fit(X, delta[:, d], iters=iters)
        return self

    def predict(self, obs: torch.Tensor, act: torch.Tensor, obs_var: Optional[torch.Tensor] = None):
        X = torch.cat([obs, act], dim=-1)
        means, variances = [], []
        for gp in self.gps:
            m, v = gp.predict(X)
            if obs_var is not None:
                # inflate predictive variance by the input uncertainty
                # through the kernel lengthscales (mean-term of the
                # moment-matching equations; cross-covariances omitted)
                ls = gp.log_lengthscale.exp()[: self.obs_dim]
                diag = obs_var.diagonal(dim1=-2, dim2=-1) if obs_var.dim() > obs.dim() else obs_var
                v = v + (diag / ls.pow(2)).sum(-1) * v
            means.append(m)
            variances.append(v)
        mu = torch.stack(means, dim=-1)
        var = torch.stack(variances, dim=-1)
        return obs + mu, var

    def forward(self, td):
        obs = td.get(("observation", "mean"), None)
        if obs is None:
            obs = td.get("observation")
            obs_var = None
        else:
            obs_var = td.get(("observation", "var"), None)
        act = td.get(("action", "mean"), None)
        if act is None:
            act = td.get("action")
        mean, var = self.predict(obs, act, obs_var)
        td.set(("next", "observation", "mean"), mean)
        td.set(("next", "observation", "var"), torch.diag_embed(var))
        return td


class RBFController(nn.Module):
    """RBF-network policy ``u = sum_i w_i k(x, c_i)`` squashed into
    action bounds (PILCO's controller)."""

    def __init__(
        self,
        obs_dim: int,
        action_dim: int,
        num_centers: int = 20,
        max_action: float = 1.0,
    ):
        super().__init__()
        self.centers = nn.Parameter(torch.randn(num_centers, obs_dim))
        self.log_lengthscale = nn.Parameter(torch.zeros(obs_dim))
        self.weights = nn.Parameter(torch.randn(num_centers, action_dim) * 0.1)
        self.max_action = max_action

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        ls = self.log_lengthscale.exp()
        d = (obs.unsqueeze(-2) / ls - self.centers / ls).pow(2).sum(-1)
        k = torch.exp(-0.5 * d)  # [..., num_centers]
        u = k @ self.weights
        return self.max_action * torch.tanh(u)
