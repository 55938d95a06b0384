"""Classification: LogisticRegression (+ RandomForestClassifier).

LogisticRegression (reference classification.py:822,1306) — distributed
L-BFGS (+OWL-QN for L1) mirroring the reference's LogisticRegressionMG call
(reference classification.py:1046-1081: linesearch_max_iter=20,
lbfgs_memory=10, penalty_normalized=False): the per-iteration work is the
fused score/softmax/gradient pass (MFMA GEMMs + fused residual kernel on
gfx950), followed by ONE all-reduce of the [C×(d+1) gradient | loss] buffer;
the optimizer itself runs replicated and bit-identical on every rank.

Spark objective implemented:
    min  1/n Σ_i logloss_i + λ[(1-α)/2 Σ p_j² + α Σ |p_j|]
where p_j is the standardized-space coefficient when standardization=True,
or the raw coefficient when False (Spark's behavior); intercept unpenalized.
Features are internally scaled by 1/σ (no centering — Spark's
sparsity-preserving choice); coefficients are returned unscaled (the
reference achieves the same via cupy standardization + un-scaling,
classification.py:1018-1028, 1127-1147).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch

from ..core import Estimator, Model, _FitContext
from ..data import to_device_tensor
from ..params import (
    HasEnableSparseDataOptim,
    HasFeaturesCol,
    HasFeaturesCols,
    HasLabelCol,
    HasPredictionCol,
    HasProbabilityCol,
    HasRawPredictionCol,
    Param,
    TypeConverters,
    HasWeightCol,
)
from ..ops.glm import logistic_grad_loss
from ..ops.lbfgs import lbfgs
from ..utils import as_numpy
from .tree import _RandomForestEstimator, _RandomForestModel


class _LogisticRegressionParams(
    HasFeaturesCol,
    HasFeaturesCols,
    HasLabelCol,
    HasPredictionCol,
    HasProbabilityCol,
    HasRawPredictionCol,
    HasEnableSparseDataOptim,
    HasWeightCol,
):
    regParam = Param("logreg", "regParam", "regularization λ.", TypeConverters.toFloat)
    elasticNetParam = Param(
        "logreg", "elasticNetParam", "L1 ratio α in [0,1].", TypeConverters.toFloat
    )
    maxIter = Param("logreg", "maxIter", "max L-BFGS iterations.", TypeConverters.toInt)
    tol = Param("logreg", "tol", "convergence tolerance.", TypeConverters.toFloat)
    fitIntercept = Param("logreg", "fitIntercept", "fit intercept.", TypeConverters.toBoolean)
    standardization = Param(
        "logreg", "standardization", "penalize in standardized space.", TypeConverters.toBoolean
    )
    family = Param(
        "logreg", "family", "auto|binomial|multinomial.", TypeConverters.toString
    )
    threshold = Param("logreg", "threshold", "binary decision threshold.", TypeConverters.toFloat)
    thresholds = Param(
        "logreg", "thresholds", "per-class thresholds (unsupported, reference classification.py:689).",
        TypeConverters.toListFloat,
    )
    weightCol = Param("logreg", "weightCol", "unsupported on GPU.", TypeConverters.toString)
    lowerBoundsOnCoefficients = Param("logreg", "lowerBoundsOnCoefficients", "unsupported.", TypeConverters.identity)
    upperBoundsOnCoefficients = Param("logreg", "upperBoundsOnCoefficients", "unsupported.", TypeConverters.identity)
    lowerBoundsOnIntercepts = Param("logreg", "lowerBoundsOnIntercepts", "unsupported.", TypeConverters.identity)
    upperBoundsOnIntercepts = Param("logreg", "upperBoundsOnIntercepts", "unsupported.", TypeConverters.identity)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(
            regParam=0.0,
            elasticNetParam=0.0,
            maxIter=100,
            tol=1e-6,
            fitIntercept=True,
            standardization=True,
            family="auto",
            threshold=0.5,
        )

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        # reference classification.py:721-760; regParam -> C inversion done in fit
        return {
            "regParam": "C",
            "elasticNetParam": "l1_ratio",
            "maxIter": "max_iter",
            "tol": "tol",
            "fitIntercept": "fit_intercept",
            "standardization": "standardization",
            "family": "",
            "threshold": "",
            "thresholds": None,
            "weightCol": None,
            "lowerBoundsOnCoefficients": None,
            "upperBoundsOnCoefficients": None,
            "lowerBoundsOnIntercepts": None,
            "upperBoundsOnIntercepts": None,
        }

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        return {
            "C": 0.0,
            "l1_ratio": 0.0,
            "max_iter": 100,
            "tol": 1e-6,
            "fit_intercept": True,
            "standardization": True,
            "penalty": "l2",
            "lbfgs_memory": 10,
            "linesearch_max_iter": 20,
            "verbose": False,
        }


class LogisticRegression(_LogisticRegressionParams, Estimator):
    """Distributed logistic regression (reference LogisticRegression,
    classification.py:822)."""

    _supports_sparse = True  # CSR path (reference classification.py:960-966)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._set_params(**kwargs)

    def setRegParam(self, value: float) -> "LogisticRegression":
        return self._set_params(regParam=value)

    def setElasticNetParam(self, value: float) -> "LogisticRegression":
        return self._set_params(elasticNetParam=value)

    def setMaxIter(self, value: int) -> "LogisticRegression":
        return self._set_params(maxIter=value)

    def setTol(self, value: float) -> "LogisticRegression":
        return self._set_params(tol=value)

    def setFitIntercept(self, value: bool) -> "LogisticRegression":
        return self._set_params(fitIntercept=value)

    def setStandardization(self, value: bool) -> "LogisticRegression":
        return self._set_params(standardization=value)

    def setFamily(self, value: str) -> "LogisticRegression":
        return self._set_params(family=value)

    def setFeaturesCol(self, value) -> "LogisticRegression":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setLabelCol(self, value: str) -> "LogisticRegression":
        return self._set_params(labelCol=value)

    def setPredictionCol(self, value: str) -> "LogisticRegression":
        return self._set_params(predictionCol=value)

    def setProbabilityCol(self, value: str) -> "LogisticRegression":
        return self._set_params(probabilityCol=value)

    # -- fit ---------------------------------------------------------------
    def _fit_array(
        self, X: Any, y: Optional[Any], ctx: _FitContext, params: Dict[str, Any]
    ) -> Dict[str, Any]:
        comm, pdesc = ctx.comm, ctx.pdesc
        # Per-param-map values land under the native keys 'C'/'l1_ratio'
        # (mapped from regParam/elasticNetParam); the native solver takes λ
        # directly, so 'C' carries λ semantics here (reference translates
        # regParam->C=1/regParam for cuML, classification.py:721-744).
        lam = float(params.get("C", self.getOrDefault("regParam")))
        l1r = float(params.get("l1_ratio", self.getOrDefault("elasticNetParam")))
        max_iter = int(params["max_iter"])
        tol = float(params["tol"])
        fit_intercept = bool(params["fit_intercept"])
        standardization = bool(params.get("standardization", True))
        family = self.getOrDefault("family")

        n = pdesc.m
        d = pdesc.n
        from ..data import _is_sparse

        sparse = _is_sparse(X)
        stream_cap: Optional[int] = None  # set when the dense shard streams
        # enable_sparse_data_optim (reference params.py:45-66): None=auto,
        # False=densify, True=require sparse input
        opt = (
            self.getOrDefault("enable_sparse_data_optim")
            if self.isSet("enable_sparse_data_optim")
            else None
        )
        if opt is False and sparse:
            X = X.toarray()
            sparse = False
        elif opt is True and not sparse:
            raise ValueError(
                "enable_sparse_data_optim=True requires sparse CSR features"
            )
        if sparse:
            # CSR path (reference classification.py:960-966: int64 index
            # escape only above 1e9 nnz; torch CSR uses int64 throughout)
            import scipy.sparse as sp

            Xcsr = X.tocsr()
            # int32 indices below the 2^31 nnz escape hatch (reference
            # classification.py:960-966,1054-1055), int64 above
            idx_t = np.int32 if Xcsr.nnz < 2**31 else np.int64
            Xt = torch.sparse_csr_tensor(
                torch.from_numpy(np.ascontiguousarray(Xcsr.indptr, dtype=idx_t)),
                torch.from_numpy(np.ascontiguousarray(Xcsr.indices, dtype=idx_t)),
                torch.from_numpy(np.ascontiguousarray(Xcsr.data, dtype=np.float32)),
                size=Xcsr.shape,
            ).to(ctx.device)
        else:
            from ..streaming import should_stream, stream_cap_bytes

            Xnp = np.ascontiguousarray(X)
            if should_stream(Xnp.nbytes, ctx.device):
                # shard exceeds the device-data cap: keep X host-resident and
                # stream chunks per L-BFGS iteration (grad accumulation on
                # device; reference capacity mechanism utils.py:403-522)
                stream_cap = stream_cap_bytes(ctx.device)
                Xt = None
            else:
                Xt = ctx.device_tensor(X)

        # classes: global sorted unique labels (reference allgathers classes_)
        local_classes = np.unique(np.asarray(y)) if len(np.asarray(y)) else np.array([])
        all_classes = comm.allgather_obj(local_classes.tolist())
        classes = np.array(sorted({c for part in all_classes for c in part}))
        n_classes = len(classes)
        if n_classes == 1:
            # single-class degenerate fit (reference classification.py:1106-1121)
            only = float(classes[0])
            if only not in (0.0, 1.0):
                raise RuntimeError(
                    "Labels MUST be in [0, 1) when the dataset has a single class"
                )
            coef = np.zeros((1, d))
            intercept = np.array([float("inf") if only == 1.0 else float("-inf")])
            return {
                "coef_": coef,
                "intercept_": intercept,
                "classes_": np.array([0.0, 1.0]),
                "n_iter_": 0,
                "objective_": 0.0,
                "num_iters_hist_": [],
            }

        # vectorized label -> class-index mapping (a python dict loop costs
        # ~0.3 µs/row — seconds at 50M rows)
        y_idx = np.searchsorted(classes, np.asarray(y)).astype(np.int64)
        y_t = to_device_tensor(y_idx, ctx.device)

        multinomial = (family == "multinomial") or (family == "auto" and n_classes > 2)
        C_out = n_classes if multinomial else 1

        # feature scale (no centering: Spark's sparsity-preserving scaling)
        if stream_cap is not None:
            from ..streaming import iter_device_chunks

            sbuf = torch.zeros((2, d), dtype=torch.float64, device=ctx.device)
            for _s, _e, Xc in iter_device_chunks(Xnp, ctx.device, stream_cap):
                sbuf[0] += Xc.sum(dim=0).to(torch.float64)
                sbuf[1] += (Xc * Xc).sum(dim=0).to(torch.float64)
            sbuf = comm.allreduce_t(sbuf)
            smean = sbuf[0] / n
            svar = torch.clamp((sbuf[1] - n * smean * smean) / max(1, n - 1), min=0.0)
            sigma = torch.sqrt(svar)
        else:
            sigma = self._column_std(Xt, comm, n)
        sig_safe = torch.where(sigma > 0, sigma, torch.ones_like(sigma))
        XsT = None
        if stream_cap is not None:
            Xs = None
        elif sparse:
            inv = (1.0 / sig_safe).to(torch.float32)
            vals = Xt.values() * inv[Xt.col_indices()]
            Xs = torch.sparse_csr_tensor(
                Xt.crow_indices(), Xt.col_indices(), vals, size=Xt.shape
            )
            from ..ops.dispatch import has_hip_ops, use_hip as _use_hip

            if _use_hip(vals) and has_hip_ops() and C_out <= 32:
                # own csr_grad kernel scatters by column — no transposed
                # CSR needed (the 400M-pair transpose sort cost ~0.5 s)
                XsT = None
            else:
                # pre-transpose once for the per-iteration XᵀR SpMM
                XsT = Xs.t().to_sparse_csr()
        else:
            Xs = Xt / sig_safe[None, :].to(Xt.dtype)

        ncol = d + (1 if fit_intercept else 0)
        # optimizer state lives on CPU: the weight vector is C*(d+1) floats,
        # so the L-BFGS two-loop recursion is hundreds of tiny ops that cost
        # more as device launches + .item() syncs than as CPU math. Only the
        # data-sized work (scores/residual/gradient GEMMs) runs on device.
        w0 = torch.zeros(C_out * ncol, dtype=torch.float64)

        pen_scale = torch.ones(d, dtype=torch.float64)
        if not standardization:
            pen_scale = 1.0 / (sig_safe.to(torch.float64).cpu() ** 2)

        l2 = lam * (1.0 - l1r)
        l1 = lam * l1r

        def _streamed_grad_loss(W: torch.Tensor):
            from ..streaming import iter_device_chunks

            inv = (1.0 / sig_safe).to(torch.float32)
            gacc = None
            lacc = torch.zeros((), dtype=torch.float64, device=ctx.device)
            for _s, _e, Xc in iter_device_chunks(
                Xnp, ctx.device, stream_cap, dtype=torch.float32
            ):
                Xcs = Xc * inv[None, :]
                g, l = logistic_grad_loss(Xcs, y_t[_s:_e], W, fit_intercept)
                gacc = g.to(torch.float64) if gacc is None else gacc + g.to(torch.float64)
                lacc += l.to(torch.float64)
            return gacc, lacc

        def closure(wv: torch.Tensor) -> Tuple[float, torch.Tensor]:
            W = wv.to(ctx.device).view(C_out, ncol).to(
                torch.float32 if (sparse or stream_cap is not None) else Xs.dtype
            )
            if stream_cap is not None:
                grad, loss = _streamed_grad_loss(W)
            else:
                grad, loss = logistic_grad_loss(Xs, y_t, W, fit_intercept, XT=XsT)
            buf = torch.zeros(C_out * ncol + 1, dtype=torch.float64, device=grad.device)
            buf[:-1] = grad.to(torch.float64).flatten()
            buf[-1] = loss.to(torch.float64)
            buf = comm.allreduce_t(buf).cpu()
            g = buf[:-1] / n
            total_loss = float(buf[-1]) / n
            if l2 > 0:
                coef = wv.view(C_out, ncol)[:, :d]
                g = g.view(C_out, ncol)
                g[:, :d] += l2 * coef * pen_scale[None, :]
                total_loss += 0.5 * l2 * float(((coef**2) * pen_scale[None, :]).sum())
                g = g.reshape(-1)
            return total_loss, g

        l1_vec = None
        if l1 > 0:
            l1_vec = torch.zeros(C_out, ncol, dtype=torch.float64)
            if standardization:
                l1_vec[:, :d] = l1
            else:
                l1_vec[:, :d] = l1 / sig_safe.to(torch.float64).cpu()[None, :]
            l1_vec = l1_vec.flatten()

        w_opt, obj, n_iter, obj_history = lbfgs(
            w0,
            closure,
            max_iter=max_iter,
            tol=tol,
            history=int(params.get("lbfgs_memory", 10)),
            l1_strength=l1_vec,
            linesearch_max_iter=int(params.get("linesearch_max_iter", 20)),
        )

        W = w_opt.view(C_out, ncol)
        coef_s = W[:, :d]
        coef = (coef_s / sig_safe.to(torch.float64).cpu()[None, :]).numpy()
        if fit_intercept:
            intercept = W[:, d].cpu().numpy()
        else:
            intercept = np.zeros(C_out)
        if multinomial:
            # center intercepts (reference classification.py:1135-1147)
            intercept = intercept - intercept.mean()

        return {
            "coef_": coef,
            "intercept_": intercept,
            "classes_": classes.astype(np.float64),
            "n_iter_": n_iter,
            "objective_": float(obj),
            "objective_history_": np.asarray(obj_history, dtype=np.float64),
        }

    def _column_std(self, Xt: torch.Tensor, comm, n: int) -> torch.Tensor:
        buf = torch.zeros((2, Xt.shape[1]), dtype=torch.float64, device=Xt.device)
        if Xt.shape[0] > 0:
            if Xt.layout == torch.sparse_csr:
                from ..ops.dispatch import has_hip_ops, hip_ops, use_hip as _uh

                vals32 = Xt.values()
                if (
                    _uh(vals32)
                    and has_hip_ops()
                    and vals32.dtype == torch.float32
                    and Xt.shape[1] * 16 <= 160 * 1024
                ):
                    # one-pass LDS-privatized column moments (torch's f64
                    # index_add over 400M nnz was the sparse-fit wall)
                    buf += hip_ops().csr_col_moments(
                        Xt.col_indices(), vals32, Xt.shape[1]
                    )
                else:
                    vals = vals32.to(torch.float64)
                    cols = Xt.col_indices()
                    buf[0].index_add_(0, cols, vals)
                    buf[1].index_add_(0, cols, vals * vals)
            else:
                # chunked, f32 partial sums accumulated in f64 (the f64-upcast
                # reduce kernel is ~40x slower; per-chunk f32 sums match the
                # reference's f32 cupy standardization precision)
                for s0 in range(0, Xt.shape[0], 1 << 18):
                    xb = Xt[s0 : s0 + (1 << 18)]
                    buf[0] += xb.sum(dim=0).to(torch.float64)
                    buf[1] += (xb * xb).sum(dim=0).to(torch.float64)
        buf = comm.allreduce_t(buf)
        mean = buf[0] / n
        var = torch.clamp((buf[1] - n * mean * mean) / max(1, n - 1), min=0.0)
        return torch.sqrt(var)

    def _create_model(self, attrs: Dict[str, Any]) -> "LogisticRegressionModel":
        return LogisticRegressionModel(**attrs)


class LogisticRegressionModel(_LogisticRegressionParams, Model):
    """Fitted logistic regression (reference LogisticRegressionModel,
    classification.py:1306). Transform computes decision_function locally —
    no comms (reference classification.py:1455-1553)."""

    def __init__(
        self,
        coef_: np.ndarray,
        intercept_: np.ndarray,
        classes_: np.ndarray,
        n_iter_: int = 0,
        objective_: float = 0.0,
        objective_history_: Optional[np.ndarray] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(
            coef_=np.asarray(coef_),
            intercept_=np.asarray(intercept_),
            classes_=np.asarray(classes_),
            n_iter_=int(n_iter_),
            objective_=float(objective_),
            objective_history_=np.asarray(
                objective_history_ if objective_history_ is not None else []
            ),
        )

    @property
    def objectiveHistory(self) -> np.ndarray:
        """Per-iteration objective (Spark LogisticRegressionTrainingSummary
        parity; reference compares full objectives,
        tests_large/test_large_logistic_regression.py:39-60)."""
        return self._model_attributes["objective_history_"]

    @property
    def coefficients(self) -> np.ndarray:
        c = self._model_attributes["coef_"]
        return c[0] if c.shape[0] == 1 else c

    @property
    def coefficientMatrix(self) -> np.ndarray:
        return self._model_attributes["coef_"]

    @property
    def intercept(self) -> float:
        i = self._model_attributes["intercept_"]
        return float(i[0]) if len(i) == 1 else float("nan")

    @property
    def interceptVector(self) -> np.ndarray:
        return self._model_attributes["intercept_"]

    @property
    def classes_(self) -> np.ndarray:
        return self._model_attributes["classes_"]

    @property
    def numClasses(self) -> int:
        return len(self.classes_)

    @property
    def numFeatures(self) -> int:
        return self._model_attributes["coef_"].shape[1]

    @property
    def coef_(self) -> np.ndarray:
        """cuml-style coefficient matrix (reference exposes the native model
        attributes alongside the Spark API)."""
        return self._model_attributes["coef_"]

    @property
    def intercept_(self) -> np.ndarray:
        return self._model_attributes["intercept_"]

    @property
    def n_cols(self) -> int:
        return self.numFeatures

    def predict(self, value) -> float:
        """Single-vector prediction (pyspark LogisticRegressionModel.predict)."""
        out = self._transform_array(np.asarray(value, dtype=np.float64).reshape(1, -1))
        return float(out[self.getOrDefault("predictionCol")][0])

    def predictRaw(self, value) -> np.ndarray:
        out = self._transform_array(np.asarray(value, dtype=np.float64).reshape(1, -1))
        return np.asarray(out[self.getOrDefault("rawPredictionCol")][0])

    def predictProbability(self, value) -> np.ndarray:
        out = self._transform_array(np.asarray(value, dtype=np.float64).reshape(1, -1))
        return np.asarray(out[self.getOrDefault("probabilityCol")][0])

    def setRawPredictionCol(self, value: str) -> "LogisticRegressionModel":
        return self._set_params(rawPredictionCol=value)

    def cpu(self):
        """Fitted sklearn.linear_model.LogisticRegression equivalent."""
        from sklearn.linear_model import LogisticRegression as SkLogReg

        sk = SkLogReg()
        sk.coef_ = np.asarray(self._model_attributes["coef_"], dtype=np.float64)
        sk.intercept_ = np.asarray(self._model_attributes["intercept_"], dtype=np.float64)
        sk.classes_ = np.asarray(self.classes_)
        sk.n_features_in_ = sk.coef_.shape[1]
        return sk

    @property
    def hasSummary(self) -> bool:
        """True when the fit recorded a training summary (Spark parity)."""
        return len(self._model_attributes.get("objective_history_", [])) > 0

    @property
    def summary(self) -> "LogisticRegressionTrainingSummary":
        if not self.hasSummary:
            raise RuntimeError("No training summary available on this model")
        hist = np.asarray(self._model_attributes["objective_history_"])
        return LogisticRegressionTrainingSummary(
            objectiveHistory=hist, totalIterations=len(hist), accuracy=None
        )

    def evaluate(self, df) -> "LogisticRegressionTrainingSummary":
        """Classification metrics on a dataset (Spark model.evaluate parity)."""
        from ..evaluation import MulticlassClassificationEvaluator

        ev = MulticlassClassificationEvaluator(
            labelCol=self.getOrDefault("labelCol"),
            predictionCol=self.getOrDefault("predictionCol"),
            metricName="accuracy",
        )
        acc = ev.evaluate(self.transform(df))
        hist = np.asarray(self._model_attributes.get("objective_history_", []))
        return LogisticRegressionTrainingSummary(
            objectiveHistory=hist, totalIterations=len(hist), accuracy=acc
        )

    def setFeaturesCol(self, value) -> "LogisticRegressionModel":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(featuresCol=value)

    def setPredictionCol(self, value: str) -> "LogisticRegressionModel":
        return self._set_params(predictionCol=value)

    def setProbabilityCol(self, value: str) -> "LogisticRegressionModel":
        return self._set_params(probabilityCol=value)

    def setThreshold(self, value: float) -> "LogisticRegressionModel":
        return self._set_params(threshold=value)

    def _transform_array(self, X: Any) -> Dict[str, np.ndarray]:
        from ..data import _is_sparse
        from ..parallel.context import get_comm

        device = get_comm().device
        if _is_sparse(X):
            Xc = X.tocsr()
            Xt = torch.sparse_csr_tensor(
                torch.from_numpy(Xc.indptr.astype(np.int64)),
                torch.from_numpy(Xc.indices.astype(np.int64)),
                torch.from_numpy(Xc.data.astype(np.float32)),
                size=Xc.shape,
            ).to(device)
        else:
            Xt = to_device_tensor(np.ascontiguousarray(X), device)
        coef = torch.from_numpy(
            np.ascontiguousarray(self._model_attributes["coef_"])
        ).to(device, Xt.dtype)
        icpt = torch.from_numpy(
            np.ascontiguousarray(self._model_attributes["intercept_"])
        ).to(device, Xt.dtype)
        # degenerate single-class model: ±inf intercept
        if bool(torch.isinf(icpt).any()):
            n = Xt.shape[0]
            pred_cls = 1.0 if float(icpt[0]) > 0 else 0.0
            probs = np.zeros((n, 2), dtype=np.float64)
            probs[:, int(pred_cls)] = 1.0
            raw = np.full((n, 2), 0.0)
            raw[:, int(pred_cls)] = np.inf
            raw[:, 1 - int(pred_cls)] = -np.inf
            return {
                self.getOrDefault("predictionCol"): np.full(n, pred_cls),
                self.getOrDefault("probabilityCol"): probs,
                self.getOrDefault("rawPredictionCol"): raw,
            }

        margins = Xt @ coef.T + icpt[None, :]
        if coef.shape[0] == 1:
            m = margins[:, 0]
            raw = torch.stack([-m, m], dim=1)
            p1 = torch.sigmoid(m)
            probs = torch.stack([1 - p1, p1], dim=1)
            thr = float(self.getOrDefault("threshold"))
            pred_idx = (p1 > thr).to(torch.int64)
        else:
            raw = margins
            probs = torch.softmax(margins, dim=1)
            pred_idx = probs.argmax(dim=1)
        classes = torch.from_numpy(
            np.ascontiguousarray(self.classes_)
        ).to(device, torch.float64)
        pred = classes[pred_idx]
        return {
            self.getOrDefault("predictionCol"): as_numpy(pred),
            self.getOrDefault("probabilityCol"): as_numpy(probs.to(torch.float64)),
            self.getOrDefault("rawPredictionCol"): as_numpy(raw.to(torch.float64)),
        }


# ---------------------------------------------------------------------------
# Random forest classification
# ---------------------------------------------------------------------------


class _RFClassifierParams(HasProbabilityCol, HasRawPredictionCol):
    pass


class RandomForestClassifier(_RFClassifierParams, _RandomForestEstimator):
    """Random forest classifier (reference classification.py:318)."""

    _task = "classification"

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._setDefault(featureSubsetStrategy="auto", impurity="gini")
        self._set_params(**kwargs)

    def setProbabilityCol(self, value: str) -> "RandomForestClassifier":
        return self._set_params(probabilityCol=value)

    def _create_model(self, attrs: Dict[str, Any]) -> "RandomForestClassificationModel":
        return RandomForestClassificationModel(**attrs)


class RandomForestClassificationModel(_RFClassifierParams, _RandomForestModel):
    """Fitted RF classification model (reference classification.py:534)."""

    _task = "classification"

    @property
    def hasSummary(self) -> bool:
        """RF fits keep no training summary (matching pyspark models loaded
        from storage); use evaluate(df) for dataset metrics."""
        return False

    def evaluate(self, df):
        from ..evaluation import MulticlassClassificationEvaluator

        ev = MulticlassClassificationEvaluator(
            labelCol=self.getOrDefault("labelCol"),
            predictionCol=self.getOrDefault("predictionCol"),
            metricName="accuracy",
        )
        return ev.evaluate(self.transform(df))


class LogisticRegressionTrainingSummary:
    """Spark LogisticRegressionTrainingSummary equivalent (training objective
    trace; accuracy filled when produced by model.evaluate)."""

    def __init__(self, objectiveHistory, totalIterations: int, accuracy=None):
        self.objectiveHistory = objectiveHistory
        self.totalIterations = totalIterations
        self.accuracy = accuracy
