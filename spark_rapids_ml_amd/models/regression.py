"""Regression: LinearRegression (OLS / Ridge / ElasticNet) + RandomForestRegressor.

LinearRegression (reference regression.py:284,698) — solver dispatch by
regularization (reference regression.py:549-606): every path starts from ONE
fused all-reduce of the sufficient statistics
[X^T X (d×d, MFMA Gram kernel) | X^T y | colsum(X) | sum(y), sum(y²), n],
after which each rank holds the full moment set and solves locally and
deterministically:

- regParam=0  -> OLS normal equations (reference LinearRegressionMG eig path)
- elasticNetParam=0 -> Ridge closed form with Spark's scaling semantics:
  the penalty applies to standardized-space coefficients with the label-std
  factor (reference scales cuML's alpha ×m and /stddev_label to match Spark,
  regression.py:573-582 — here the Spark objective is implemented natively)
- otherwise -> coordinate descent on the GRAM (covariance) formulation: the
  reference's CDMG does a per-sweep all-reduce over the data
  (SURVEY.md §2.3b); because the d×d Gram is already replicated, CD here
  needs NO further communication — glmnet-style covariance updates.

Spark objective implemented (fitIntercept, standardization=True):
    min_w̃  1/(2n)||ỹ - X̃ w̃||² + λ[(1-α)/2 ||w̃||² + α ||w̃||₁]
with X̃ column-standardized, ỹ = (y-ȳ)/σ_y, coefficients returned in raw
space (w_j = w̃_j σ_y/σ_xj). With standardization=False the penalty is on
w̃_j/σ_xj (raw-coefficient penalization, Spark's behavior).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch

from ..core import Estimator, Model, _FitContext
from ..data import to_device_tensor
from ..params import (
    HasFeaturesCol,
    HasFeaturesCols,
    HasLabelCol,
    HasPredictionCol,
    Param,
    TypeConverters,
    HasWeightCol,
)
from ..ops import gram
from ..utils import as_numpy
from .tree import _RandomForestEstimator, _RandomForestModel


class _LinearRegressionParams(HasFeaturesCol, HasFeaturesCols, HasLabelCol, HasPredictionCol, HasWeightCol):
    regParam = Param("linreg", "regParam", "regularization strength λ.", TypeConverters.toFloat)
    elasticNetParam = Param(
        "linreg", "elasticNetParam", "L1 ratio α in [0,1].", TypeConverters.toFloat
    )
    fitIntercept = Param("linreg", "fitIntercept", "fit intercept.", TypeConverters.toBoolean)
    standardization = Param(
        "linreg", "standardization", "penalize in standardized space.", TypeConverters.toBoolean
    )
    maxIter = Param("linreg", "maxIter", "max CD sweeps.", TypeConverters.toInt)
    tol = Param("linreg", "tol", "CD convergence tolerance.", TypeConverters.toFloat)
    solver = Param("linreg", "solver", "auto|normal|eig.", TypeConverters.toString)
    loss = Param("linreg", "loss", "squaredError only.", TypeConverters.toString)
    weightCol = Param("linreg", "weightCol", "unsupported on GPU.", TypeConverters.toString)
    aggregationDepth = Param("linreg", "aggregationDepth", "ignored.", TypeConverters.toInt)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(
            regParam=0.0,
            elasticNetParam=0.0,
            fitIntercept=True,
            standardization=True,
            maxIter=100,
            tol=1e-6,
            solver="auto",
            loss="squaredError",
            aggregationDepth=2,
        )

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        # reference regression.py:183-215
        return {
            "regParam": "alpha",
            "elasticNetParam": "l1_ratio",
            "fitIntercept": "fit_intercept",
            "standardization": "normalize",
            "maxIter": "max_iter",
            "tol": "tol",
            "solver": "solver",
            "loss": "loss",
            "weightCol": None,
            "aggregationDepth": "",
        }

    @classmethod
    def _param_value_mapping(cls):
        return {
            "loss": lambda v: "squared_loss" if v in ("squaredError", "squared_loss") else None,
            "solver": lambda v: {"auto": "eig", "normal": "eig", "eig": "eig"}.get(v, None),
        }

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        return {
            "alpha": 0.0,
            "l1_ratio": 0.0,
            "fit_intercept": True,
            "normalize": True,
            "max_iter": 100,
            "tol": 1e-6,
            "solver": "eig",
            "loss": "squared_loss",
            # cuML-signature knobs accepted for ctor parity (inert here)
            "algorithm": "auto",
            "copy_X": True,
            "shuffle": True,
            "verbose": False,
        }


class LinearRegression(_LinearRegressionParams, Estimator):
    """Distributed linear regression (reference LinearRegression,
    regression.py:284)."""

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._set_params(**kwargs)

    def setTol(self, value: float) -> "LinearRegression":
        return self._set_params(tol=value)

    def setSolver(self, value: str) -> "LinearRegression":
        return self._set_params(solver=value)

    def setLoss(self, value: str) -> "LinearRegression":
        # reference supports squaredError only on GPU (regression.py:240)
        if value != "squaredError":
            raise ValueError("Only loss='squaredError' is supported.")
        return self._set_params(loss=value)

    def setRegParam(self, value: float) -> "LinearRegression":
        return self._set_params(regParam=value)

    def setElasticNetParam(self, value: float) -> "LinearRegression":
        return self._set_params(elasticNetParam=value)

    def setMaxIter(self, value: int) -> "LinearRegression":
        return self._set_params(maxIter=value)

    def setFitIntercept(self, value: bool) -> "LinearRegression":
        return self._set_params(fitIntercept=value)

    def setStandardization(self, value: bool) -> "LinearRegression":
        return self._set_params(standardization=value)

    def setFeaturesCol(self, value) -> "LinearRegression":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setLabelCol(self, value: str) -> "LinearRegression":
        return self._set_params(labelCol=value)

    def setPredictionCol(self, value: str) -> "LinearRegression":
        return self._set_params(predictionCol=value)

    # -- fit ---------------------------------------------------------------
    def _moments(self, X, y, ctx: _FitContext) -> Dict[str, torch.Tensor]:
        """One data pass + ONE fused all-reduce -> replicated sufficient
        statistics. Cached across param maps of a fitMultiple sweep."""
        if "moments" in ctx.cache:
            return ctx.cache["moments"]
        comm, pdesc = ctx.comm, ctx.pdesc
        d = pdesc.n
        from ..streaming import should_stream, stream_cap_bytes, streamed_moments

        Xnp = np.asarray(X) if not hasattr(X, "toarray") else X
        # fused buffer rows: [0:d]=Gram, d=X^T y, d+1=colsum X, d+2=[ysum, y2sum]
        # (width >= 2 so the label-moment row exists even at d=1)
        w = max(d, 2)
        if (
            isinstance(Xnp, np.ndarray)
            and should_stream(Xnp.nbytes, ctx.device)
            and Xnp.shape[0] > 0
        ):
            # shard exceeds the device-data cap: pinned-chunk streaming pass
            mom = streamed_moments(
                Xnp, np.asarray(y), ctx.device, stream_cap_bytes(ctx.device)
            )
            buf = torch.zeros((d + 3, w), dtype=torch.float64, device=ctx.device)
            buf[:d, :d] = mom["G"]
            buf[d, :d] = mom["Xty"]
            buf[d + 1, :d] = mom["xsum"]
            buf[d + 2, 0] = mom["ysum"]
            buf[d + 2, 1] = mom["y2sum"]
        else:
            Xt = to_device_tensor(np.ascontiguousarray(X), ctx.device)
            yt = to_device_tensor(np.ascontiguousarray(y), ctx.device).to(Xt.dtype)
            buf = torch.zeros((d + 3, w), dtype=torch.float64, device=Xt.device)
            if Xt.shape[0] > 0:
                buf[:d, :d] = gram(Xt).to(torch.float64)
                buf[d, :d] = (Xt.T @ yt).to(torch.float64)
                buf[d + 1, :d] = Xt.sum(dim=0).to(torch.float64)
                buf[d + 2, 0] = yt.sum().to(torch.float64)
                buf[d + 2, 1] = (yt * yt).sum().to(torch.float64)
        buf = comm.allreduce_t(buf)
        m = {
            "G": buf[:d, :d],
            "Xty": buf[d, :d],
            "xsum": buf[d + 1, :d],
            "ysum": buf[d + 2, 0],
            "y2sum": buf[d + 2, 1],
            "n": pdesc.m,
        }
        ctx.cache["moments"] = m
        return m

    def _fit_array(
        self, X: Any, y: Optional[Any], ctx: _FitContext, params: Dict[str, Any]
    ) -> Dict[str, Any]:
        mom = self._moments(X, y, ctx)
        n = mom["n"]
        d = ctx.pdesc.n
        lam = float(params["alpha"])
        l1r = float(params["l1_ratio"])
        fit_intercept = bool(params["fit_intercept"])
        standardization = bool(params.get("normalize", True))
        max_iter = int(params["max_iter"])
        tol = float(params["tol"])

        G, Xty, xsum, ysum, y2sum = (
            mom["G"],
            mom["Xty"],
            mom["xsum"],
            mom["ysum"],
            mom["y2sum"],
        )
        xbar = xsum / n
        ybar = ysum / n if fit_intercept else torch.zeros_like(ysum)

        # centered second moments
        if fit_intercept:
            Gc = G - n * torch.outer(xbar, xbar)
            bc = Xty - n * xbar * ybar
            yvar = torch.clamp((y2sum - n * ybar * ybar) / max(1, n - 1), min=0.0)
        else:
            Gc = G
            bc = Xty
            yvar = torch.clamp(y2sum / max(1, n - 1), min=0.0)
        xvar = torch.clamp(torch.diagonal(Gc) / max(1, n - 1), min=0.0)
        sx = torch.sqrt(xvar)
        sy = torch.sqrt(yvar)
        sx_safe = torch.where(sx > 0, sx, torch.ones_like(sx))
        sy_safe = sy if float(sy) > 0 else torch.ones_like(sy)

        if lam == 0.0:
            # OLS: solve Gc w = bc (jittered Cholesky, pinv fallback)
            w = _sym_solve(Gc, bc)
        else:
            # standardized-space problem: Q = X̃ᵀX̃/n, r = X̃ᵀỹ/n
            denom = torch.outer(sx_safe, sx_safe) * n
            Q = Gc / denom
            r = bc / (sx_safe * sy_safe * n)
            if standardization:
                pen_scale = torch.ones_like(sx)
            else:
                pen_scale = 1.0 / torch.clamp(sx_safe, min=1e-30) ** 2
            if l1r == 0.0:
                A = Q + lam * torch.diag(pen_scale)
                wt = _sym_solve(A, r)
            else:
                wt = _cd_elasticnet(
                    Q, r, lam, l1r, pen_scale, max_iter=max_iter, tol=tol
                )
            w = wt * sy_safe / sx_safe  # unscale (reference regression.py:634-648)
            w = torch.where(sx > 0, w, torch.zeros_like(w))

        intercept = float((ybar - (w * xbar).sum()).item()) if fit_intercept else 0.0

        # training objective value (for summary parity)
        w64 = w.to(torch.float64)
        sse = float(
            (
                y2sum
                - 2.0 * (w64 * Xty).sum()
                + w64 @ G @ w64
                + (
                    2.0 * intercept * ((w64 * xsum).sum() - ysum)
                    + n * intercept * intercept
                    if fit_intercept
                    else 0.0
                )
            ).item()
        )

        # total sum of squares around the label mean (for summary r2)
        tss = float((y2sum - ysum * ysum / max(1, n)).item())
        return {
            "coef_": as_numpy(w).astype(np.float64),
            "intercept_": intercept,
            "n_rows_": n,
            "sse_": max(0.0, sse),
            "tss_": max(0.0, tss),
        }

    def _create_model(self, attrs: Dict[str, Any]) -> "LinearRegressionModel":
        return LinearRegressionModel(**attrs)


def _sym_solve(A: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Solve symmetric PSD system with jittered Cholesky, lstsq fallback
    (reference LinearRegressionMG 'eig' path solves via eigendecomposition)."""
    A = A.to(torch.float64)
    b = b.to(torch.float64)
    d = A.shape[0]
    jitter = 1e-12 * torch.clamp(torch.diagonal(A).mean(), min=1e-30)
    for _ in range(6):
        try:
            L = torch.linalg.cholesky(A + jitter * torch.eye(d, dtype=A.dtype, device=A.device))
            return torch.cholesky_solve(b.view(-1, 1), L).flatten()
        except Exception:
            jitter = jitter * 100 + 1e-10
    return torch.linalg.lstsq(A.cpu(), b.view(-1, 1).cpu()).solution.flatten().to(A.device)


def _cd_elasticnet(
    Q: torch.Tensor,
    r: torch.Tensor,
    lam: float,
    l1r: float,
    pen_scale: torch.Tensor,
    max_iter: int,
    tol: float,
) -> torch.Tensor:
    """Coordinate descent on the Gram formulation (communication-free:
    Q = X̃ᵀX̃/n and r = X̃ᵀỹ/n are already replicated).

    obj(w) = 1/2 wᵀQw - rᵀw + λ Σ_j s_j[(1-α)/2 w_j² + α|w_j|]
    update: w_j <- S(r_j - Σ_{i≠j} Q_ij w_i, λ α s_j) / (Q_jj + λ(1-α) s_j)

    Runs on numpy: the replicated d×d Gram is tiny and the per-coordinate
    loop would otherwise pay a device sync per coordinate.
    """
    import numpy as _np

    dev = Q.device
    Qn = Q.detach().cpu().to(torch.float64).numpy()
    rn = r.detach().cpu().to(torch.float64).numpy()
    pn = pen_scale.detach().cpu().to(torch.float64).numpy()
    d = Qn.shape[0]
    w = _np.zeros(d)
    qdiag = _np.diagonal(Qn).copy()
    denom = _np.maximum(qdiag + lam * (1.0 - l1r) * pn, 1e-30)
    thresh = lam * l1r * pn
    Qw = _np.zeros(d)
    for sweep in range(max_iter):
        w_max = 0.0
        d_max = 0.0
        for j in range(d):
            wj_old = w[j]
            rho = rn[j] - Qw[j] + qdiag[j] * wj_old
            t = thresh[j]
            if rho > t:
                wj = (rho - t) / denom[j]
            elif rho < -t:
                wj = (rho + t) / denom[j]
            else:
                wj = 0.0
            diff = wj - wj_old
            if diff != 0.0:
                Qw += Qn[:, j] * diff
                w[j] = wj
            w_max = max(w_max, abs(wj))
            d_max = max(d_max, abs(diff))
        if w_max == 0.0 or d_max / max(w_max, 1e-30) < tol:
            break
    return torch.from_numpy(w).to(dev)


class LinearRegressionModel(_LinearRegressionParams, Model):
    """Fitted linear regression model (reference LinearRegressionModel,
    regression.py:698)."""

    def __init__(
        self,
        coef_: np.ndarray,
        intercept_: float = 0.0,
        n_rows_: int = 0,
        sse_: float = 0.0,
        tss_: float = 0.0,
        **kwargs: Any,
    ) -> None:
        super().__init__(
            coef_=np.asarray(coef_),
            intercept_=float(intercept_),
            n_rows_=int(n_rows_),
            sse_=float(sse_),
            tss_=float(tss_),
        )

    @property
    def coefficients(self) -> np.ndarray:
        return self._model_attributes["coef_"]

    @property
    def coef_(self) -> np.ndarray:
        return self._model_attributes["coef_"]

    @property
    def intercept(self) -> float:
        return self._model_attributes["intercept_"]

    @property
    def numFeatures(self) -> int:
        return len(self.coefficients)

    @property
    def hasSummary(self) -> bool:
        return self._model_attributes.get("n_rows_", 0) > 0

    @property
    def summary(self) -> "LinearRegressionTrainingSummary":
        """Training summary (Spark LinearRegressionTrainingSummary parity:
        rootMeanSquaredError/meanSquaredError from the training SSE)."""
        n = max(1, self._model_attributes["n_rows_"])
        sse = self._model_attributes["sse_"]
        tss = self._model_attributes.get("tss_", 0.0)
        return LinearRegressionTrainingSummary(
            meanSquaredError=sse / n,
            rootMeanSquaredError=float(np.sqrt(sse / n)),
            numInstances=n,
            r2=(1.0 - sse / tss) if tss > 0 else float("nan"),
        )

    def cpu(self):
        """Fitted sklearn.linear_model.LinearRegression equivalent."""
        from sklearn.linear_model import LinearRegression as SkLR

        sk = SkLR()
        sk.coef_ = np.asarray(self.coefficients, dtype=np.float64)
        sk.intercept_ = float(self.intercept)
        sk.n_features_in_ = len(sk.coef_)
        return sk

    @property
    def scale(self) -> float:
        """Spark LinearRegressionModel.scale — 1.0 except for huber loss,
        which neither this build nor the reference supports."""
        return 1.0

    def evaluate(self, df) -> "LinearRegressionTrainingSummary":
        """Regression metrics on a dataset (Spark model.evaluate parity):
        transforms locally, merges sufficient statistics across ranks."""
        from ..evaluation import RegressionEvaluator

        ev = RegressionEvaluator(
            labelCol=self.getOrDefault("labelCol"),
            predictionCol=self.getOrDefault("predictionCol"),
        )
        from ..parallel.context import get_comm

        out = self.transform(df)
        mse = ev.setMetricName("mse").evaluate(out)
        n = int(get_comm().allreduce_scalar(float(len(df))))
        return LinearRegressionTrainingSummary(
            meanSquaredError=mse,
            rootMeanSquaredError=float(np.sqrt(mse)),
            numInstances=n,
        )

    def setFeaturesCol(self, value) -> "LinearRegressionModel":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setPredictionCol(self, value: str) -> "LinearRegressionModel":
        return self._set_params(predictionCol=value)

    def predict(self, vector: np.ndarray) -> float:
        return float(np.dot(np.asarray(vector), self.coefficients) + self.intercept)

    def _transform_array(self, X: Any) -> np.ndarray:
        from ..parallel.context import get_comm

        device = get_comm().device
        Xt = to_device_tensor(np.ascontiguousarray(X), device)
        w = torch.from_numpy(np.ascontiguousarray(self.coefficients)).to(device, Xt.dtype)
        pred = Xt @ w + self.intercept
        return as_numpy(pred).astype(np.float64)


# ---------------------------------------------------------------------------
# Random forest regression (shared machinery in tree.py)
# ---------------------------------------------------------------------------


class RandomForestRegressor(_RandomForestEstimator):
    """Random forest regressor (reference regression.py:880): trees are
    split across ranks, each rank fits on its local shard only — no
    collectives during fit (reference tree.py:330-341)."""

    _task = "regression"

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._setDefault(featureSubsetStrategy="auto", impurity="variance")
        self._set_params(**kwargs)

    def _create_model(self, attrs: Dict[str, Any]) -> "RandomForestRegressionModel":
        return RandomForestRegressionModel(**attrs)


class RandomForestRegressionModel(_RandomForestModel):
    """Fitted RF regression model (reference regression.py:1055)."""

    _task = "regression"


class LinearRegressionTrainingSummary:
    """Spark LinearRegressionTrainingSummary equivalent (training-set
    metrics from the fit's sufficient statistics)."""

    def __init__(
        self,
        meanSquaredError: float,
        rootMeanSquaredError: float,
        numInstances: int,
        r2: float = float("nan"),
    ):
        self.meanSquaredError = meanSquaredError
        self.rootMeanSquaredError = rootMeanSquaredError
        self.numInstances = numInstances
        self.r2 = r2
