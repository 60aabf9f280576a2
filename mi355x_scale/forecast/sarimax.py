"""SARIMAX-lite: regression-with-ARMA-errors, Hannan–Rissanen estimator.

The reference fits statsmodels ``SARIMAX(endog, exog, order=(p,d,q))``
per SKU with a Nelder-Mead/Kalman loop
(``group_apply/02_Fine_Grained_Demand_Forecasting.py:441-450,472-481``) —
data-dependent iteration counts, hostile to batched GPU execution
(SURVEY §7 hard-part 1). This module fixes the estimator to a closed-form
friendly pipeline with a FIXED op schedule:

  1. difference the series d times (exog too),
  2. OLS of w on [1, X]  → regression residuals u,
  3. long-AR OLS on u    → innovation estimates ê,
  4. OLS of u on [u lags, ê lags] → (φ, θ),
  5. one refinement pass: recompute ê from (φ,θ), redo step 4.

Every step is dense linear algebra of identical shape for every group —
exactly what the batched CDNA4 kernel (``forecast/batched.py``) runs; this
CPU implementation is its numerics reference, and the statistical-parity
target is forecast MSE on the seed-123 data (not bitwise-vs-statsmodels).

API shape follows statsmodels' SARIMAX the way the reference calls it:
``SARIMAX(y, exog=X, order=(p,d,q)).fit()`` → results with
``.predict(start, end, exog=...)``, ``.fittedvalues``, ``.resid``, ``.mse``.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional, Tuple

import numpy as np


def _ols(X: np.ndarray, y: np.ndarray) -> np.ndarray:
    """Least squares via normal equations with ridge jitter (the exact
    computation the GPU kernel performs: XtX, Xty, Cholesky solve)."""
    XtX = X.T @ X
    XtX[np.diag_indices_from(XtX)] += 1e-8 * max(1.0, np.trace(XtX) / len(XtX))
    Xty = X.T @ y
    return np.linalg.solve(XtX, Xty)


def _difference(y: np.ndarray, d: int) -> np.ndarray:
    for _ in range(d):
        y = np.diff(y)
    return y


def _build_lag_matrix(u: np.ndarray, e: np.ndarray, p: int, q: int,
                      m: int) -> Tuple[np.ndarray, np.ndarray]:
    """Rows t = m..T-1: [u_{t-1..t-p}, e_{t-1..t-q}] → target u_t."""
    T = len(u)
    rows = T - m
    X = np.empty((rows, p + q))
    for i in range(p):
        X[:, i] = u[m - 1 - i:T - 1 - i]
    for j in range(q):
        X[:, p + j] = e[m - 1 - j:T - 1 - j]
    return X, u[m:]


@dataclass
class SARIMAXResults:
    order: Tuple[int, int, int]
    const: float
    beta: np.ndarray          # exog coefficients
    phi: np.ndarray           # AR coefficients (p)
    theta: np.ndarray         # MA coefficients (q)
    endog: np.ndarray         # original series
    exog: Optional[np.ndarray]
    w: np.ndarray             # differenced series
    u: np.ndarray             # regression residuals (differenced scale)
    eps: np.ndarray           # innovation estimates (aligned with u)
    fittedvalues: np.ndarray = field(init=False)
    resid: np.ndarray = field(init=False)

    def __post_init__(self):
        self.fittedvalues = self._insample_fit()
        self.resid = self.endog - self.fittedvalues

    # -- in-sample ---------------------------------------------------------
    def _what(self) -> np.ndarray:
        """One-step-ahead fitted values of the differenced series."""
        p, q = len(self.phi), len(self.theta)
        T = len(self.w)
        reg = np.full(T, self.const)
        if self.exog is not None and self.beta.size:
            Xd = _difference_cols(self.exog, self.order[1])
            reg = reg + Xd @ self.beta
        what = reg.copy()
        for t in range(T):
            acc = 0.0
            for i in range(p):
                if t - 1 - i >= 0:
                    acc += self.phi[i] * self.u[t - 1 - i]
            for j in range(q):
                if t - 1 - j >= 0:
                    acc += self.theta[j] * self.eps[t - 1 - j]
            what[t] += acc
        return what

    def _insample_fit(self) -> np.ndarray:
        d = self.order[1]
        what = self._what()
        if d == 0:
            return what
        # undifference: ŷ_t = y_{t-1} (+ y diffs) + ŵ; one-step-ahead uses
        # observed history.
        y = self.endog
        fit = np.empty_like(y, dtype=np.float64)
        fit[:d] = y[:d]
        if d == 1:
            fit[1:] = y[:-1] + what
        else:  # d == 2
            fit[2:] = 2 * y[1:-1] - y[:-2] + what
        return fit

    @property
    def mse(self) -> float:
        d = self.order[1]
        r = self.resid[d + max(len(self.phi), len(self.theta)):]
        return float(np.mean(r * r)) if len(r) else float("inf")

    @property
    def params(self) -> np.ndarray:
        return np.concatenate([[self.const], self.beta, self.phi, self.theta])

    # -- forecasting ---------------------------------------------------------
    def forecast(self, steps: int,
                 exog: Optional[np.ndarray] = None) -> np.ndarray:
        """Out-of-sample forecast ``steps`` ahead (exog: [steps, k])."""
        p, q, d = len(self.phi), len(self.theta), self.order[1]
        u_hist = list(self.u)
        e_hist = list(self.eps)
        if exog is not None and self.beta.size:
            exf = np.asarray(exog, dtype=np.float64)
            # difference the future exog against the tail of the history
            full = np.vstack([self.exog, exf]) if self.exog is not None else exf
            Xd = _difference_cols(full, d)[-steps:]
        else:
            Xd = None
        w_fore = np.empty(steps)
        for h in range(steps):
            reg = self.const + (Xd[h] @ self.beta if Xd is not None else 0.0)
            acc = 0.0
            for i in range(p):
                k = len(u_hist) - 1 - i
                if k >= 0:
                    acc += self.phi[i] * u_hist[k]
            for j in range(q):
                k = len(e_hist) - 1 - j
                if k >= 0:
                    acc += self.theta[j] * e_hist[k]
            w_fore[h] = reg + acc
            u_hist.append(acc)    # future u = ARMA part (E[e]=0)
            e_hist.append(0.0)
        # integrate back
        y = self.endog
        if d == 0:
            return w_fore
        if d == 1:
            return y[-1] + np.cumsum(w_fore)
        out = np.empty(steps)
        y1, y2 = y[-1], y[-2]
        for h in range(steps):
            nxt = w_fore[h] + 2 * y1 - y2
            out[h] = nxt
            y2, y1 = y1, nxt
        return out

    def predict(self, start: int = 0, end: Optional[int] = None,
                exog: Optional[np.ndarray] = None) -> np.ndarray:
        """statsmodels-style: in-sample one-step-ahead for t < T, dynamic
        forecast beyond (reference use: ``group_apply/02_...py:484-488``)."""
        T = len(self.endog)
        if end is None:
            end = T - 1
        n_out = end - T + 1
        vals = list(self.fittedvalues)
        if n_out > 0:
            vals += list(self.forecast(n_out, exog=exog))
        return np.array(vals[start:end + 1])


def _difference_cols(X: np.ndarray, d: int) -> np.ndarray:
    X = np.asarray(X, dtype=np.float64)
    if X.ndim == 1:
        X = X[:, None]
    for _ in range(d):
        X = np.diff(X, axis=0)
    return X


class SARIMAX:
    """statsmodels-shaped constructor; extra statsmodels kwargs accepted
    and ignored (enforce_stationarity etc., ref ``:447-448``)."""

    LONG_AR_EXTRA = 3

    def __init__(self, endog, exog=None, order=(1, 0, 0), **_ignored):
        self.endog = np.asarray(endog, dtype=np.float64).ravel()
        self.exog = (np.asarray(exog, dtype=np.float64)
                     if exog is not None else None)
        if self.exog is not None and self.exog.ndim == 1:
            self.exog = self.exog[:, None]
        p, d, q = order
        if d > 2:
            raise ValueError("d <= 2 supported")
        self.order = (int(p), int(d), int(q))

    def fit(self, disp: bool = False, method: str = "hr",
            refine: int = 1, **_ignored) -> SARIMAXResults:
        p, d, q = self.order
        w = _difference(self.endog, d)
        T = len(w)
        min_rows = p + q + (1 if self.exog is None else 1 + self.exog.shape[1])
        if T < max(8, min_rows + 4):
            raise ValueError(f"series too short for order {self.order}")
        # 1) regression part
        if self.exog is not None:
            Xd = _difference_cols(self.exog, d)
            Xr = np.hstack([np.ones((T, 1)), Xd])
        else:
            Xr = np.ones((T, 1))
        coef = _ols(Xr, w)
        const, beta = float(coef[0]), coef[1:]
        u = w - Xr @ coef

        # 2) long-AR for innovations
        if q > 0:
            m_ar = min(max(p, q) + self.LONG_AR_EXTRA, max(1, T // 4))
            Xa = np.column_stack([u[m_ar - 1 - i:T - 1 - i]
                                  for i in range(m_ar)])
            ya = u[m_ar:]
            a = _ols(Xa, ya)
            eps = np.zeros(T)
            eps[m_ar:] = ya - Xa @ a
        else:
            eps = u.copy()

        phi = np.zeros(p)
        theta = np.zeros(q)
        if p + q > 0:
            m = max(p, q, 1)
            for _ in range(1 + max(0, refine)):
                X2, y2 = _build_lag_matrix(u, eps, p, q, m)
                c2 = _ols(X2, y2)
                phi, theta = c2[:p], c2[p:]
                # Shrink toward stationarity/invertibility: the ê and
                # forecast recursions diverge when Σ|φ| or Σ|θ| > 1
                # (statsmodels hides this inside enforce_stationarity;
                # here the fixed-schedule estimator clips instead — same
                # rule as the GPU kernel).
                sp = np.sum(np.abs(phi))
                if sp > 0.98:
                    phi *= 0.98 / sp
                st = np.sum(np.abs(theta))
                if st > 0.98:
                    theta *= 0.98 / st
                # recompute innovations under (phi, theta)
                new_eps = np.zeros(T)
                for t in range(T):
                    acc = u[t]
                    for i in range(p):
                        if t - 1 - i >= 0:
                            acc -= phi[i] * u[t - 1 - i]
                    for j in range(q):
                        if t - 1 - j >= 0:
                            acc -= theta[j] * new_eps[t - 1 - j]
                    new_eps[t] = acc
                eps = new_eps
                if q == 0:
                    break

        return SARIMAXResults(
            order=self.order, const=const, beta=beta, phi=phi, theta=theta,
            endog=self.endog, exog=self.exog, w=w, u=u, eps=eps)
