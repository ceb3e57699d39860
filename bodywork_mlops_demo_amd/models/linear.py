"""Closed-form OLS linear regressor on MI355X.

The reference trains sklearn ``LinearRegression`` (LAPACK lstsq under the
hood, ``stage_1_train_model.py:105-106``).  Here the fit is one fused HIP
statistics reduction (``ops.linreg_stats``) + a host-side 2x2 closed-form
solve, and in data-parallel training the five fp64 statistics are the
entire RCCL all-reduce payload (latency-bound — SURVEY.md §5).

Artefact format: the persisted joblib object is a *real sklearn
``LinearRegression``* populated with the GPU-computed coefficients, so a
reference user can ``joblib.load`` and ``predict`` with nothing but
sklearn installed (artefact-parity with ``stage_1:111-125``).
"""
from __future__ import annotations

import torch

from bodywork_mlops_demo_amd.utils.device import canonical_device

from bodywork_mlops_demo_amd import ops


class GPULinearRegressor:
    """y = intercept + coef * x, fit by fused-stats closed form."""

    def __init__(self, intercept: float = 0.0, coef: float = 0.0, device="cpu"):
        self.intercept_ = float(intercept)
        self.coef_ = float(coef)
        self.device = canonical_device(device)
        self._ab: torch.Tensor | None = None  # device-resident [a, b]

    def _ab_tensor(self) -> torch.Tensor:
        """2-element device tensor read by the scoring kernel — captured
        graphs follow weight updates through this buffer (no recapture)."""
        if self._ab is None or self._ab.device != self.device:
            self._ab = torch.tensor([self.intercept_, self.coef_],
                                    device=self.device, dtype=torch.float32)
        return self._ab

    def _sync_ab(self) -> None:
        if self._ab is not None:
            self._ab.copy_(torch.tensor([self.intercept_, self.coef_],
                                        dtype=torch.float32))

    def copy_weights_from(self, other: "GPULinearRegressor") -> bool:
        """In-place weight swap (serving hot-redeploy path)."""
        if not isinstance(other, GPULinearRegressor):
            return False
        self.intercept_, self.coef_ = other.intercept_, other.coef_
        self._sync_ab()
        return True

    # -- training ----------------------------------------------------------
    def fit(self, X: torch.Tensor, y: torch.Tensor, process_group=None):
        """Fit on (possibly sharded) data.

        With ``process_group`` set, each rank passes its shard; the fused
        statistics are summed with one all-reduce before the solve, which
        is numerically identical to a single-GPU fit up to fp64 reduction
        order.
        """
        stats = ops.linreg_stats(X, y)
        if process_group is not None:
            import torch.distributed as dist

            dist.all_reduce(stats, group=process_group)
        self.intercept_, self.coef_ = ops.solve_ols(stats.cpu())
        self._sync_ab()
        return self

    # -- inference ---------------------------------------------------------
    def predict(self, X: torch.Tensor) -> torch.Tensor:
        if X.device.type == "cuda":
            return ops.linear_score(X, ab=self._ab_tensor())
        return ops.linear_score(X, self.intercept_, self.coef_)

    def to(self, device):
        self.device = canonical_device(device)
        self._ab = None
        return self

    # -- artefact compatibility --------------------------------------------
    def to_sklearn(self):
        import numpy as np
        from sklearn.linear_model import LinearRegression

        m = LinearRegression(fit_intercept=True)
        m.coef_ = np.array([self.coef_])
        m.intercept_ = float(self.intercept_)
        m.n_features_in_ = 1
        return m

    @classmethod
    def from_sklearn(cls, m, device="cpu") -> "GPULinearRegressor":
        return cls(float(m.intercept_), float(m.coef_.ravel()[0]), device)

    def __repr__(self) -> str:  # str(model) appears in the scoring response
        return "LinearRegression()"
