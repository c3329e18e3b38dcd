"""ML_DETECT_ANOMALIES: streaming per-key forecaster with confidence bands.

Semantics contract (LAB3-Walkthrough.md:119-133, LAB4-Walkthrough.md:150-164):
``ML_DETECT_ANOMALIES(value, ts, JSON params) OVER (PARTITION BY key ORDER BY
ts)`` returns ``(forecast_value, upper_bound, lower_bound, is_anomaly)`` per
row.  Params: minTrainingSize, maxTrainingSize, confidencePercentage,
enableStl.  Until minTrainingSize history points exist, rows are never
anomalies.  The detector must reproduce the labs' determinism contracts:
lab3 (minTrain 286, conf 99.9) flags only the French Quarter surge window;
lab4 (minTrain 8, maxTrain 50, conf 95) flags only the Naples spike.

Model: AR(p) with intercept, fit by least squares on the rolling history
(ARIMA-style one-step forecast); the confidence band is the training
residual std times the two-sided normal quantile.  This CPU implementation
is the numerics reference for the batched HIP kernel in ops/hip/anomaly.hip
(one workgroup per key, normal equations in LDS).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np


def normal_quantile(p: float) -> float:
    """Two-sided -> one-sided handled by caller; Acklam/Moro-style inverse CDF."""
    # Peter Acklam's rational approximation; |rel err| < 1.15e-9.
    if not 0.0 < p < 1.0:
        raise ValueError("p in (0,1)")
    a = [-3.969683028665376e+01, 2.209460984245205e+02, -2.759285104469687e+02,
         1.383577518672690e+02, -3.066479806614716e+01, 2.506628277459239e+00]
    b = [-5.447609879822406e+01, 1.615858368580409e+02, -1.556989798598866e+02,
         6.680131188771972e+01, -1.328068155288572e+01]
    c = [-7.784894002430293e-03, -3.223964580411365e-01, -2.400758277161838e+00,
         -2.549732539343734e+00, 4.374664141464968e+00, 2.938163982698783e+00]
    d = [7.784695709041462e-03, 3.224671290700398e-01, 2.445134137142996e+00,
         3.754408661907416e+00]
    plow, phigh = 0.02425, 1 - 0.02425
    if p < plow:
        q = math.sqrt(-2 * math.log(p))
        return (((((c[0] * q + c[1]) * q + c[2]) * q + c[3]) * q + c[4]) * q + c[5]) / \
               ((((d[0] * q + d[1]) * q + d[2]) * q + d[3]) * q + 1)
    if p > phigh:
        q = math.sqrt(-2 * math.log(1 - p))
        return -(((((c[0] * q + c[1]) * q + c[2]) * q + c[3]) * q + c[4]) * q + c[5]) / \
               ((((d[0] * q + d[1]) * q + d[2]) * q + d[3]) * q + 1)
    q = p - 0.5
    r = q * q
    return (((((a[0] * r + a[1]) * r + a[2]) * r + a[3]) * r + a[4]) * r + a[5]) * q / \
           (((((b[0] * r + b[1]) * r + b[2]) * r + b[3]) * r + b[4]) * r + 1)


@dataclass
class AnomalyResult:
    forecast_value: float
    upper_bound: float
    lower_bound: float
    is_anomaly: bool


def ar_forecast(history: np.ndarray, order: int) -> tuple[float, float, int]:
    """One-step AR(p)+intercept forecast with a prediction standard error.

    p is capped so the fit keeps residual degrees of freedom (an AR(4) on 8
    points interpolates exactly and reports a zero band — the overfit trap).
    Returns (forecast, prediction_se, dof) where prediction_se includes the
    parameter-uncertainty factor sqrt(1 + x'(X'X)^+ x).
    """
    n = len(history)
    p = min(order, max(1, (n - 4) // 4)) if n >= 6 else 0
    if p < 1:
        mean = float(history.mean())
        std = float(history.std(ddof=1)) if n > 1 else abs(mean) + 1.0
        return mean, max(std, 1e-9), max(n - 1, 1)
    # Design: rows t = p..n-1 predict history[t] from lags 1..p, centered,
    # with RIDGE shrinkage on the lag coefficients: OLS on short windows
    # fits noise into recent lags and drags the one-step forecast off the
    # local level (false positives); ridge leaves strong (trend) directions
    # intact and shrinks noise-fitting directions toward mean reversion.
    m = n - p
    X = np.empty((m, p), dtype=np.float64)
    for j in range(1, p + 1):
        X[:, j - 1] = history[p - j:n - j]
    y = history[p:]
    xm = X.mean(axis=0)
    ym = float(y.mean())
    Xc = X - xm
    yc = y - ym
    G = Xc.T @ Xc
    lam = 0.3 * (np.trace(G) / p + 1e-12)
    coef = np.linalg.solve(G + lam * np.eye(p), Xc.T @ yc)
    resid = yc - Xc @ coef
    dof = max(m - (p + 1), 1)
    resid_var = float(resid @ resid) / dof
    x_next = history[n - 1:n - p - 1:-1].astype(np.float64)
    forecast = ym + float((x_next - xm) @ coef)
    # Prediction interval: widen by parameter uncertainty at x_next.
    Ginv = np.linalg.inv(G + lam * np.eye(p))
    lever = float((x_next - xm) @ Ginv @ (x_next - xm)) + 1.0 / m
    pred_se = math.sqrt(max(resid_var, 1e-18) * (1.0 + max(lever, 0.0)))
    return forecast, max(pred_se, 1e-9), dof


class AnomalyDetector:
    """Streaming per-key detector with ML_DETECT_ANOMALIES semantics."""

    DEFAULT_ORDER = 4

    def __init__(self, min_training_size: int = 10, max_training_size: int = 1000,
                 confidence_percentage: float = 99.0, enable_stl: bool = False,
                 order: int | None = None):
        self.min_training = int(min_training_size)
        self.max_training = int(max_training_size)
        self.confidence = float(confidence_percentage)
        self.enable_stl = bool(enable_stl)  # STL decomposition not modeled
        self.order = order or self.DEFAULT_ORDER
        self.z = normal_quantile(0.5 + self.confidence / 200.0)
        self._history: dict[str, list[float]] = {}

    @classmethod
    def from_json_params(cls, params: dict) -> "AnomalyDetector":
        return cls(
            min_training_size=params.get("minTrainingSize", 10),
            max_training_size=params.get("maxTrainingSize", 1000),
            confidence_percentage=params.get("confidencePercentage", 99.0),
            enable_stl=params.get("enableStl", False),
        )

    def update(self, key: str, value: float) -> AnomalyResult:
        hist = self._history.setdefault(key, [])
        if len(hist) < self.min_training:
            res = AnomalyResult(float(value), float("inf"), float("-inf"), False)
        else:
            arr = np.asarray(hist, dtype=np.float64)
            forecast, pred_se, dof = ar_forecast(arr, self.order)
            # Tolerance-interval band: (a) Cornish-Fisher normal -> Student-t
            # adjustment for the quantile; (b) Wilson-Hilferty chi^2 80% UCB
            # on the variance estimate (sigma-hat from few samples biases low,
            # which would leak false positives through an exact z-band).
            zt = self.z + (self.z ** 3 + self.z) / (4.0 * max(dof, 1))
            h = 2.0 / (9.0 * max(dof, 1))
            chi2_low = max(dof, 1) * (1.0 - h - 0.8416 * math.sqrt(h)) ** 3
            pred_se *= math.sqrt(max(dof, 1) / max(chi2_low, 1e-9))
            upper = forecast + zt * pred_se
            lower = forecast - zt * pred_se
            res = AnomalyResult(forecast, upper, lower,
                                bool(value > upper or value < lower))
        hist.append(float(value))
        if len(hist) > self.max_training:
            del hist[: len(hist) - self.max_training]
        return res

    def series_results(self, key: str, values: list[float]) -> list[AnomalyResult]:
        return [self.update(key, v) for v in values]

    # ------------------------------------------------------------------
    def batch_results_gpu(self, series_by_key: dict[str, list[float]],
                          device: str = "cuda") -> dict[str, list["AnomalyResult"]]:
        """Every (key, window-step) scored in ONE kernel launch: step i of a
        key becomes its own ragged row (the rolling history before i,
        max_training-trimmed) of the [rows, Tmax] matrix fed to the batched
        AR HIP kernel (ops/hip/streaming.hip); the tolerance-band math then
        vectorizes on host.  Mutates per-key history exactly like the
        sequential path, and matches it (GPU test asserts)."""
        import torch

        from ..ops import dispatch as D
        jobs = []          # (key, step_idx, prefix)
        out: dict[str, list[AnomalyResult]] = {}
        for key, values in series_by_key.items():
            hist = self._history.setdefault(key, [])
            out[key] = [None] * len(values)   # type: ignore[list-item]
            for i, v in enumerate(values):
                if len(hist) < self.min_training:
                    out[key][i] = AnomalyResult(float(v), float("inf"),
                                                float("-inf"), False)
                else:
                    jobs.append((key, i,
                                 np.asarray(hist[-self.max_training:],
                                            dtype=np.float32)))
                hist.append(float(v))
            if len(hist) > self.max_training:
                del hist[: len(hist) - self.max_training]
        if jobs:
            tmax = max(len(p) for _, _, p in jobs)
            mat = np.zeros((len(jobs), tmax), dtype=np.float32)
            lens = np.zeros(len(jobs), dtype=np.int32)
            for r, (_, _, p) in enumerate(jobs):
                mat[r, :len(p)] = p
                lens[r] = len(p)
            mt = torch.from_numpy(mat).to(device)
            lt = torch.from_numpy(lens).to(device)
            fc, se, dof = D.ext().anomaly_batch(mt, lt, self.order)
            fc = fc.cpu().numpy()
            se = se.cpu().numpy()
            dof = np.maximum(dof.cpu().numpy(), 1).astype(np.float64)
            zt = self.z + (self.z ** 3 + self.z) / (4.0 * dof)
            h = 2.0 / (9.0 * dof)
            chi2_low = dof * (1.0 - h - 0.8416 * np.sqrt(h)) ** 3
            se_adj = se * np.sqrt(dof / np.maximum(chi2_low, 1e-9))
            upper = fc + zt * se_adj
            lower = fc - zt * se_adj
            for r, (key, i, _) in enumerate(jobs):
                v = series_by_key[key][i]
                out[key][i] = AnomalyResult(
                    float(fc[r]), float(upper[r]), float(lower[r]),
                    bool(v > upper[r] or v < lower[r]))
        return out
