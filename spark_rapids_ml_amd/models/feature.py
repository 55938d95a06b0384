"""Feature: distributed PCA (reference feature.py:117,291).

Fit computes the global column mean and Gram partials on-device (MFMA f32
SYRK kernel on gfx950), all-reduces mean (d) + Gram (d×d) in one fused RCCL
buffer (reference PCAMG allreduces partial mean and covariance separately,
SURVEY.md §2.3b), forms the covariance, eigendecomposes, applies the
deterministic sign-flip convention (reference rapidsml_jni.cu:35-61).

Transform matches Spark semantics: projection WITHOUT mean-centering
(reference adds the mean-offset back to undo cuML's centering,
feature.py:438-449; here the projection is computed Spark's way directly).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from ..core import Estimator, Model, _FitContext
from ..data import to_device_tensor
from ..params import (
    HasFeaturesCol,
    HasFeaturesCols,
    HasOutputCol,
    Param,
    TypeConverters,
)
from ..ops import gram, eigh_sym, sign_flip
from ..ops.linalg import cov_from_gram
from ..utils import as_numpy


class _PCAParams(HasFeaturesCol, HasFeaturesCols, HasOutputCol):
    k = Param("pca", "k", "number of principal components.", TypeConverters.toInt)
    inputCol = Param("pca", "inputCol", "input column name.", TypeConverters.toString)

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self._setDefault(outputCol="pca_features")

    @classmethod
    def _param_mapping(cls) -> Dict[str, Optional[str]]:
        # reference feature.py:61-75
        return {"k": "n_components", "inputCol": "", "outputCol": ""}

    @classmethod
    def _get_native_params_default(cls) -> Dict[str, Any]:
        return {
            "n_components": None,
            "svd_solver": "full",
            "whiten": False,
            "verbose": False,
        }

    def getK(self) -> int:
        return self.getOrDefault("k")

    def _get_input_columns(self):
        if self.isSet("inputCol"):
            return self.getOrDefault("inputCol"), None
        return super()._get_input_columns()


class PCA(_PCAParams, Estimator):
    """Distributed PCA estimator (reference PCA, feature.py:117)."""

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()
        self._set_params(**kwargs)

    def setK(self, value: int) -> "PCA":
        return self._set_params(k=value)

    def setInputCol(self, value) -> "PCA":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(inputCol=value)

    def setOutputCol(self, value: str) -> "PCA":
        return self._set_params(outputCol=value)

    def setInputCols(self, value) -> "PCA":
        return self._set_params(featuresCols=list(value))

    def _fit_array(
        self, X: Any, y: Optional[Any], ctx: _FitContext, params: Dict[str, Any]
    ) -> Dict[str, Any]:
        comm, pdesc = ctx.comm, ctx.pdesc
        k = int(params["n_components"] or self.getK())
        d = pdesc.n
        if k > min(pdesc.m, d):
            raise ValueError(f"k={k} > min(n_rows={pdesc.m}, n_cols={d})")

        from ..streaming import should_stream, stream_cap_bytes, streamed_moments

        Xnp = np.asarray(X) if not hasattr(X, "toarray") else X
        # fused partials: [d+1, d] buffer = [Gram (d×d) ; colsum (1×d)]
        if (
            isinstance(Xnp, np.ndarray)
            and should_stream(Xnp.nbytes, ctx.device)
            and Xnp.shape[0] > 0
        ):
            # shard exceeds the device-data cap: stream chunks through a
            # pinned buffer, accumulating the same partials on device
            mom = streamed_moments(Xnp, None, ctx.device, stream_cap_bytes(ctx.device))
            buf = torch.zeros((d + 1, d), dtype=torch.float64, device=ctx.device)
            buf[:d] = mom["G"]
            buf[d] = mom["xsum"]
        else:
            Xt = ctx.device_tensor(X)
            buf = torch.zeros((d + 1, d), dtype=torch.float64, device=Xt.device)
            if Xt.shape[0] > 0:
                g = gram(Xt)  # f32 MFMA on GPU
                buf[:d] = g.to(torch.float64)
                buf[d] = Xt.sum(dim=0).to(torch.float64)
        buf = comm.allreduce_t(buf)
        mean = buf[d] / pdesc.m
        cov = cov_from_gram(buf[:d], mean, pdesc.m, ddof=1)

        w, v = eigh_sym(cov)  # ascending
        w = torch.flip(w, dims=[0])[:k]  # top-k eigenvalues, descending
        v = torch.flip(v, dims=[1])[:, :k]  # [d,k]
        components = sign_flip(v.T.contiguous())  # [k,d]

        total_var = float(torch.clamp(torch.diagonal(cov).sum(), min=1e-300).item())
        explained = torch.clamp(w, min=0.0)
        ratio = explained / total_var
        singular = torch.sqrt(explained * max(1, pdesc.m - 1))

        return {
            "components_": as_numpy(components).astype(np.float64),
            "explained_variance_": as_numpy(explained),
            "explained_variance_ratio_": as_numpy(ratio),
            "singular_values_": as_numpy(singular),
            "mean_": as_numpy(mean),
            "n_rows_": pdesc.m,
        }

    def _create_model(self, attrs: Dict[str, Any]) -> "PCAModel":
        return PCAModel(**attrs)


class PCAModel(_PCAParams, Model):
    """Fitted PCA model (reference PCAModel, feature.py:291)."""

    def setInputCols(self, value) -> "PCAModel":
        return self._set_params(featuresCols=list(value))

    def __init__(
        self,
        components_: np.ndarray,
        explained_variance_: np.ndarray,
        explained_variance_ratio_: np.ndarray = None,
        singular_values_: np.ndarray = None,
        mean_: np.ndarray = None,
        n_rows_: int = 0,
        **kwargs: Any,
    ) -> None:
        super().__init__(
            components_=np.asarray(components_),
            explained_variance_=np.asarray(explained_variance_),
            explained_variance_ratio_=np.asarray(
                explained_variance_ratio_
                if explained_variance_ratio_ is not None
                else explained_variance_
            ),
            singular_values_=np.asarray(
                singular_values_ if singular_values_ is not None else explained_variance_
            ),
            mean_=np.asarray(mean_ if mean_ is not None else np.zeros(np.asarray(components_).shape[1])),
            n_rows_=int(n_rows_),
        )

    @property
    def pc(self) -> np.ndarray:
        """Principal components as a d×k matrix (Spark DenseMatrix layout)."""
        return self._model_attributes["components_"].T

    @property
    def components_(self) -> np.ndarray:
        return self._model_attributes["components_"]

    @property
    def explainedVariance(self) -> np.ndarray:
        """Ratio vector, matching Spark PCAModel.explainedVariance."""
        return self._model_attributes["explained_variance_ratio_"]

    @property
    def explained_variance_(self) -> np.ndarray:
        return self._model_attributes["explained_variance_"]

    @property
    def explained_variance_ratio_(self) -> np.ndarray:
        return self._model_attributes["explained_variance_ratio_"]

    @property
    def singular_values_(self) -> np.ndarray:
        return self._model_attributes["singular_values_"]

    @property
    def mean_(self) -> np.ndarray:
        return self._model_attributes["mean_"]

    @property
    def mean(self) -> list:
        """Column means as a list (reference PCAModel.mean, feature.py:291)."""
        return self._model_attributes["mean_"].tolist()

    def cpu(self):
        """Fitted sklearn.decomposition.PCA equivalent (reference cpu()
        builds the Spark PCAModel via py4j, feature.py:375-389)."""
        from sklearn.decomposition import PCA as SkPCA

        k = self.components_.shape[0]
        sk = SkPCA(n_components=k)
        sk.components_ = np.asarray(self.components_, dtype=np.float64)
        sk.explained_variance_ = np.asarray(self.explained_variance_, dtype=np.float64)
        sk.explained_variance_ratio_ = np.asarray(self.explained_variance_ratio_, dtype=np.float64)
        sk.singular_values_ = np.asarray(self.singular_values_, dtype=np.float64)
        sk.mean_ = np.asarray(self.mean_, dtype=np.float64)
        sk.n_components_ = k
        sk.n_features_in_ = sk.components_.shape[1]
        sk.n_samples_ = int(self._model_attributes.get("n_rows_", 0))
        sk.noise_variance_ = 0.0
        return sk

    def setInputCol(self, value) -> "PCAModel":
        if isinstance(value, (list, tuple)):
            return self._set_params(featuresCols=list(value))
        return self._set_params(inputCol=value)

    def setOutputCol(self, value: str) -> "PCAModel":
        return self._set_params(outputCol=value)

    def _out_col_name(self) -> str:
        return self.getOrDefault("outputCol")

    def _transform_array(self, X: Any) -> np.ndarray:
        from ..parallel.context import get_comm

        device = get_comm().device
        Xt = to_device_tensor(
            np.ascontiguousarray(X, dtype=np.float32 if self._float32_inputs else np.float64),
            device,
        )
        P = torch.from_numpy(
            np.ascontiguousarray(self.components_.T)
        ).to(device, Xt.dtype)  # [d,k]
        out = Xt @ P  # Spark: no mean centering (reference feature.py:438-449)
        return as_numpy(out)
