"""Polynomial ridge regressor on MI355X — the third model family.

The k-feature generalisation of the closed-form OLS path (SURVEY.md §2.2
mapping table: "fused XᵀX + Xᵀy reduction ... generalizes"): the fit is
one fused HIP pass (`ops.poly_stats`) that expands the normalised
polynomial basis ``phi_j = ((x-50)/50)^j`` on the fly — the design
matrix never exists in memory — followed by a host-side
ridge-regularised normal-equation solve; scoring is a Horner kernel
whose coefficients live in a device buffer (captured serving graphs
follow redeploys, like the linear model).

Drift rationale: the generator's concept drift moves the intercept
sinusoidally; between retrains a curved fit tracks residual structure a
line cannot, and the ridge term keeps the high-order terms tame on
1440-row days.

Artefact: a stock sklearn ``Pipeline(PolynomialFeatures, Ridge)`` whose
RAW-x coefficients are computed from the normalised fit by exact fp64
change of basis — ``joblib.load`` + ``predict`` work with stock sklearn.
"""
from __future__ import annotations

import numpy as np
import torch

from bodywork_mlops_demo_amd.utils.device import canonical_device

from bodywork_mlops_demo_amd import ops


class GPUPolyRegressor:
    X_MU = 50.0
    X_SIGMA = 50.0  # scale to t in [-1, 1] for U(0,100) inputs

    def __init__(self, degree: int = 3, l2: float = 1e-6, device="cpu"):
        if not 1 <= degree <= 5:
            raise ValueError("degree must be in [1, 5]")
        self.degree = degree
        self.l2 = float(l2)
        self.device = canonical_device(device)
        self.coef_t_: list[float] = [0.0] * (degree + 1)  # normalised basis
        self._coef_dev: torch.Tensor | None = None

    # -- training ----------------------------------------------------------
    def fit(self, X: torch.Tensor, y: torch.Tensor, process_group=None):
        stats = ops.poly_stats(X, y, self.degree, self.X_MU, self.X_SIGMA)
        if process_group is not None:
            import torch.distributed as dist

            dist.all_reduce(stats, group=process_group)
        self.coef_t_ = ops.solve_poly(stats, self.degree, self.l2)
        self._sync_dev()
        return self

    # -- inference ----------------------------------------------------------
    def _coef_tensor(self) -> torch.Tensor:
        if self._coef_dev is None or self._coef_dev.device != self.device:
            self._coef_dev = torch.tensor(self.coef_t_, device=self.device,
                                          dtype=torch.float32)
        return self._coef_dev

    def _sync_dev(self) -> None:
        if self._coef_dev is not None:
            self._coef_dev.copy_(torch.tensor(self.coef_t_,
                                              dtype=torch.float32))

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        if X.device.type == "cuda":
            return ops.poly_score(X, self._coef_tensor(), self.X_MU,
                                  self.X_SIGMA)
        return ops.poly_score(X, self.coef_t_, self.X_MU, self.X_SIGMA)

    def to(self, device):
        self.device = canonical_device(device)
        self._coef_dev = None
        return self

    def copy_weights_from(self, other: "GPUPolyRegressor") -> bool:
        """In-place coefficient swap (serving hot-redeploy)."""
        if (not isinstance(other, GPUPolyRegressor)
                or other.degree != self.degree):
            return False
        self.coef_t_ = list(other.coef_t_)
        self.l2 = other.l2
        self._sync_dev()
        return True

    # -- basis change (exact, fp64) -----------------------------------------
    def _raw_coefs(self) -> np.ndarray:
        """q(x) = p((x - mu)/s) expanded in powers of x."""
        from numpy.polynomial import polynomial as P

        p = np.asarray(self.coef_t_, dtype=np.float64)
        sub = np.array([-self.X_MU / self.X_SIGMA, 1.0 / self.X_SIGMA])
        q = np.array([p[-1]])
        for c in p[-2::-1]:  # Horner in polynomial arithmetic
            q = P.polyadd(P.polymul(q, sub), [c])
        return np.pad(q, (0, self.degree + 1 - len(q)))

    @classmethod
    def _from_raw_coefs(cls, raw: np.ndarray, degree: int, l2: float,
                        device) -> "GPUPolyRegressor":
        from numpy.polynomial import polynomial as P

        sub = np.array([cls.X_MU, cls.X_SIGMA])  # x = mu + s*t
        p = np.array([raw[-1]])
        for c in raw[-2::-1]:
            p = P.polyadd(P.polymul(p, sub), [c])
        self = cls(degree=degree, l2=l2, device=device)
        self.coef_t_ = np.pad(p, (0, degree + 1 - len(p))).tolist()
        return self

    # -- artefact compatibility ---------------------------------------------
    def to_sklearn(self):
        from sklearn.linear_model import Ridge
        from sklearn.pipeline import Pipeline
        from sklearn.preprocessing import PolynomialFeatures

        raw = self._raw_coefs()
        poly = PolynomialFeatures(degree=self.degree, include_bias=False)
        poly.fit(np.zeros((1, 1)))
        ridge = Ridge(alpha=self.l2)
        ridge.coef_ = raw[1:].copy()
        ridge.intercept_ = float(raw[0])
        ridge.n_features_in_ = self.degree
        return Pipeline([("poly", poly), ("ridge", ridge)])

    @classmethod
    def from_sklearn(cls, pipe, device="cpu") -> "GPUPolyRegressor":
        ridge = pipe.named_steps["ridge"]
        degree = pipe.named_steps["poly"].degree
        raw = np.concatenate([[float(ridge.intercept_)],
                              np.asarray(ridge.coef_, dtype=np.float64)])
        return cls._from_raw_coefs(raw, degree, float(ridge.alpha), device)

    def __repr__(self) -> str:
        return (f"Pipeline(PolynomialFeatures(degree={self.degree}), "
                f"Ridge(alpha={self.l2}))")
