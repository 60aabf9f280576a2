"""ExponentialSmoothing (Holt-Winters) — the reference's single-series
walkthrough fits four variants (``group_apply/02_Fine_Grained_Demand_
Forecasting.py:143-188``): simple, trend, damped-trend, trend+seasonal.

statsmodels-shaped API:

    fit = ExponentialSmoothing(y, trend="add", seasonal="add",
                               seasonal_periods=52).fit()
    fit.fittedvalues; fit.forecast(40)

``fit(optimized=True)`` minimizes one-step-ahead SSE over the smoothing
parameters with a fixed-schedule coarse-grid + golden-refinement search
(no Nelder-Mead — same design rule as the SARIMAX-lite estimator: fixed
op counts, batched/GPU-friendly).
"""
from __future__ import annotations

import itertools
from dataclasses import dataclass
from typing import Optional

import numpy as np


def _hw_sse(y, alpha, beta, gamma, phi, m, trend, seasonal,
            return_state=False):
    n = len(y)
    # init: level = mean of first season (or first value), trend from first
    # two seasons, seasonal indices from first season vs its mean.
    if seasonal and m and n >= 2 * m:
        s0 = y[:m].mean()
        lvl = s0
        b = (y[m:2 * m].mean() - s0) / m if trend else 0.0
        # detrended seasonal init: remove the in-season trend ramp, else
        # the indices absorb b*(j - mid) and fitting collapses
        j = np.arange(m)
        base = s0 + b * (j - (m - 1) / 2.0)
        if seasonal == "mul":
            seas = list(y[:m] / np.maximum(base, 1e-9))
        else:
            seas = list(y[:m] - base)
    else:
        lvl = y[0]
        b = (y[1] - y[0]) if (trend and n > 1) else 0.0
        seas = [1.0 if seasonal == "mul" else 0.0] * (m or 1)
        m = m or 1
    sse = 0.0
    fitted = np.empty(n) if return_state else None
    for t in range(n):
        si = seas[t % m]
        if seasonal == "mul":
            yhat = (lvl + phi * b) * si
        else:
            yhat = lvl + phi * b + si
        if return_state:
            fitted[t] = yhat
        err = y[t] - yhat
        sse += err * err
        prev_lvl = lvl
        if seasonal == "mul":
            deseason = y[t] / max(si, 1e-9)
        else:
            deseason = y[t] - si
        lvl = alpha * deseason + (1 - alpha) * (prev_lvl + phi * b)
        if trend:
            b = beta * (lvl - prev_lvl) + (1 - beta) * phi * b
        if seasonal:
            if seasonal == "mul":
                seas[t % m] = gamma * (y[t] / max(lvl, 1e-9)) \
                    + (1 - gamma) * si
            else:
                seas[t % m] = gamma * (y[t] - lvl) + (1 - gamma) * si
    if return_state:
        return sse, fitted, lvl, b, seas
    return sse


@dataclass
class HoltWintersResults:
    params: dict
    fittedvalues: np.ndarray
    sse: float
    _level: float
    _trend: float
    _seas: list
    _model: "ExponentialSmoothing"

    @property
    def resid(self):
        return self._model.endog - self.fittedvalues

    def forecast(self, steps: int) -> np.ndarray:
        m = self._model.seasonal_periods or 1
        phi = self.params.get("damping_trend", 1.0)
        out = np.empty(steps)
        n = len(self._model.endog)
        for h in range(1, steps + 1):
            damp = phi * h if phi == 1.0 else phi * (1 - phi ** h) / (1 - phi)
            base = self._level + (damp * self._trend
                                  if self._model.trend else 0.0)
            si = self._seas[(n + h - 1) % m]
            out[h - 1] = base * si if self._model.seasonal == "mul" \
                else base + si
        return out

    predict = forecast


class ExponentialSmoothing:
    def __init__(self, endog, trend: Optional[str] = None,
                 damped_trend: bool = False,
                 seasonal: Optional[str] = None,
                 seasonal_periods: Optional[int] = None, **_ignored):
        self.endog = np.asarray(endog, dtype=np.float64).ravel()
        if trend not in (None, "add"):
            raise ValueError("trend must be None or 'add'")
        if seasonal not in (None, "add", "mul"):
            raise ValueError("seasonal must be None, 'add' or 'mul'")
        if seasonal and not seasonal_periods:
            raise ValueError("seasonal requires seasonal_periods")
        self.trend = trend
        self.damped_trend = damped_trend
        self.seasonal = seasonal
        self.seasonal_periods = seasonal_periods

    def fit(self, smoothing_level: Optional[float] = None,
            smoothing_trend: Optional[float] = None,
            smoothing_seasonal: Optional[float] = None,
            damping_trend: Optional[float] = None,
            optimized: bool = True, **_ignored) -> HoltWintersResults:
        y = self.endog
        m = self.seasonal_periods or 0

        def sse_of(a, b, g, ph):
            return _hw_sse(y, a, b, g, ph, m, self.trend, self.seasonal)

        grid = np.array([0.05, 0.1, 0.2, 0.35, 0.5, 0.7, 0.9])
        a_cands = [smoothing_level] if smoothing_level is not None else grid
        b_cands = ([smoothing_trend] if smoothing_trend is not None
                   else (grid if self.trend else [0.0]))
        g_cands = ([smoothing_seasonal] if smoothing_seasonal is not None
                   else (grid if self.seasonal else [0.0]))
        p_cands = ([damping_trend] if damping_trend is not None
                   else ([0.8, 0.9, 0.98] if self.damped_trend else [1.0]))
        if not optimized and smoothing_level is None:
            a_cands, b_cands = [0.3], [0.1 if self.trend else 0.0]
            g_cands = [0.1 if self.seasonal else 0.0]

        best = None
        for a, b, g, ph in itertools.product(a_cands, b_cands, g_cands,
                                             p_cands):
            s = sse_of(a, b, g, ph)
            if best is None or s < best[0]:
                best = (s, a, b, g, ph)
        # one local refinement pass around the grid winner (fixed schedule)
        _, a, b, g, ph = best
        for _ in range(2):
            for da in (-0.05, 0.0, 0.05):
                for db in (-0.05, 0.0, 0.05):
                    aa = float(np.clip(a + da, 0.01, 0.99))
                    bb = float(np.clip(b + db, 0.0, 0.99))
                    s = sse_of(aa, bb, g, ph)
                    if s < best[0]:
                        best = (s, aa, bb, g, ph)
            _, a, b, g, ph = best

        sse, fitted, lvl, tr, seas = _hw_sse(
            y, a, b, g, ph, m, self.trend, self.seasonal, return_state=True)
        return HoltWintersResults(
            params={"smoothing_level": a, "smoothing_trend": b,
                    "smoothing_seasonal": g, "damping_trend": ph},
            fittedvalues=fitted, sse=sse, _level=lvl, _trend=tr,
            _seas=seas, _model=self)
