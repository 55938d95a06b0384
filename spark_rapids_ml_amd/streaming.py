"""Chunked host→pinned→HBM streaming for fits larger than the device cap.

The reference reserves a fraction of GPU memory for data and streams Arrow
batches through it (reference utils.py:403-522, conf
`spark.rapids.ml.gpuMemRatioForData`). Here the same conf
(`gpu_mem_ratio_for_data`, env SRML_GPU_MEM_RATIO_FOR_DATA) bounds the
device-resident DATA bytes: a moment-space fit (PCA, linear regression —
their sufficient statistics are one pass) or an iterative GLM (logistic
regression — one pass per L-BFGS iteration) whose shard exceeds the cap
streams row chunks through ONE reused pinned staging buffer instead of
materializing the shard in HBM; accumulation (Gram, column sums, gradients)
stays on device in f64. Results are bit-comparable to the in-memory path up
to float summation order.

SRML_STREAM_CAP_BYTES overrides the cap with an absolute byte count (used by
tests and CPU runs, where there is no device total to take a ratio of).
"""

from __future__ import annotations

import os
from typing import Iterator, Optional, Tuple

import numpy as np
import torch

from .config import get_conf


def stream_cap_bytes(device: torch.device) -> Optional[int]:
    """The device-data byte cap, or None when streaming is not configured."""
    env = os.environ.get("SRML_STREAM_CAP_BYTES")
    if env:
        return int(env)
    ratio = get_conf("gpu_mem_ratio_for_data")
    if ratio and device.type == "cuda" and torch.cuda.is_available():
        total = torch.cuda.get_device_properties(device).total_memory
        return int(float(ratio) * total)
    return None


def should_stream(nbytes: int, device: torch.device) -> bool:
    cap = stream_cap_bytes(device)
    return cap is not None and nbytes > cap


def iter_device_chunks(
    X: np.ndarray,
    device: torch.device,
    cap_bytes: int,
    dtype: torch.dtype = None,
) -> Iterator[Tuple[int, int, torch.Tensor]]:
    """Yield (start, end, device_chunk) row slices of host array X.

    Chunks are sized to a quarter of the cap (chunk + accumulators + the
    consumer's temporaries stay under it) and staged through one reused
    pinned buffer so the H2D copy runs at pinned bandwidth."""
    n, d = X.shape
    itemsize = X.dtype.itemsize
    rows = max(1, int((cap_bytes // 4) // max(1, d * itemsize)))
    rows = min(rows, n)
    use_pin = device.type == "cuda" and torch.cuda.is_available()
    if use_pin:
        tdtype = torch.from_numpy(np.empty(0, dtype=X.dtype)).dtype
        # TWO pinned buffers alternated, each guarded by an event recorded
        # after its async H2D: the host must not rewrite a pinned buffer the
        # DMA may still be reading
        pins = [torch.empty((rows, d), dtype=tdtype, pin_memory=True) for _ in range(2)]
        evs = [torch.cuda.Event() for _ in range(2)]
        recorded = [False, False]
    for it, s in enumerate(range(0, n, rows)):
        e = min(n, s + rows)
        chunk = np.ascontiguousarray(X[s:e])
        if use_pin:
            b = it & 1
            if recorded[b]:
                evs[b].synchronize()
            pv = pins[b][: e - s]
            pv.copy_(torch.from_numpy(chunk))
            t = pv.to(device, non_blocking=True)
            evs[b].record()
            recorded[b] = True
        else:
            t = torch.from_numpy(chunk).to(device)
        if dtype is not None:
            t = t.to(dtype)
        yield s, e, t


def streamed_moments(
    X: np.ndarray,
    y: Optional[np.ndarray],
    device: torch.device,
    cap_bytes: int,
) -> dict:
    """One streamed pass -> f64 device accumulators: G = XᵀX [d,d],
    xsum [d], and with y also Xty [d], ysum, y2sum (the moment set PCA and
    linear regression fit from; reference PCAMG / LinearRegressionMG
    partials)."""
    from .ops import gram

    d = X.shape[1]
    G = torch.zeros((d, d), dtype=torch.float64, device=device)
    xsum = torch.zeros(d, dtype=torch.float64, device=device)
    Xty = torch.zeros(d, dtype=torch.float64, device=device)
    ysum = torch.zeros((), dtype=torch.float64, device=device)
    y2sum = torch.zeros((), dtype=torch.float64, device=device)
    for s, e, Xc in iter_device_chunks(X, device, cap_bytes):
        G += gram(Xc).to(torch.float64)
        xsum += Xc.sum(dim=0).to(torch.float64)
        if y is not None:
            yc = torch.from_numpy(np.ascontiguousarray(y[s:e])).to(device).to(Xc.dtype)
            Xty += (Xc.T @ yc).to(torch.float64)
            ysum += yc.sum().to(torch.float64)
            y2sum += (yc * yc).sum().to(torch.float64)
    return {"G": G, "xsum": xsum, "Xty": Xty, "ysum": ysum, "y2sum": y2sum}
