"""Search-space DSL — the ``hp.*`` / ``scope.int`` surface.

Covers the expression types the reference actually uses
(``group_apply/02_Fine_Grained_Demand_Forecasting.py:291-295`` —
``scope.int(hp.quniform(...))``; ``hyperopt/1. hyperopt.py:72`` —
``hp.lognormal``; ``hyperopt/2. hyperopt on diff sizes of data.py:52`` —
``hp.uniform``) plus the common extras (loguniform, normal, choice,
randint).

Each node can sample from its prior and convert to/from an unconstrained
internal coordinate used by the TPE kernel-density machinery.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any, Dict, Sequence

import numpy as np


class Expr:
    label: str

    def sample(self, rng: np.random.Generator):
        raise NotImplementedError

    # internal (unbounded-ish) coordinate for the Parzen estimator
    def to_internal(self, value) -> float:
        return float(value)

    def from_internal(self, z: float):
        return z

    def clip_internal(self, z: float) -> float:
        return z


@dataclass
class Uniform(Expr):
    label: str
    low: float
    high: float

    def sample(self, rng):
        return float(rng.uniform(self.low, self.high))

    def clip_internal(self, z):
        return float(min(max(z, self.low), self.high))


@dataclass
class QUniform(Expr):
    label: str
    low: float
    high: float
    q: float

    def sample(self, rng):
        v = rng.uniform(self.low, self.high)
        return float(np.round(v / self.q) * self.q)

    def from_internal(self, z):
        return float(np.round(z / self.q) * self.q)

    def clip_internal(self, z):
        return float(min(max(z, self.low), self.high))


@dataclass
class LogUniform(Expr):
    label: str
    low: float   # log-space bounds, like hyperopt: value = exp(U(low, high))
    high: float

    def sample(self, rng):
        return float(math.exp(rng.uniform(self.low, self.high)))

    def to_internal(self, value):
        return math.log(value)

    def from_internal(self, z):
        return float(math.exp(z))

    def clip_internal(self, z):
        return float(min(max(z, self.low), self.high))


@dataclass
class Normal(Expr):
    label: str
    mu: float
    sigma: float

    def sample(self, rng):
        return float(rng.normal(self.mu, self.sigma))


@dataclass
class LogNormal(Expr):
    label: str
    mu: float
    sigma: float

    def sample(self, rng):
        return float(math.exp(rng.normal(self.mu, self.sigma)))

    def to_internal(self, value):
        return math.log(max(value, 1e-300))

    def from_internal(self, z):
        return float(math.exp(z))


@dataclass
class QLogUniform(Expr):
    label: str
    low: float
    high: float
    q: float

    def sample(self, rng):
        v = math.exp(rng.uniform(self.low, self.high))
        return float(max(np.round(v / self.q) * self.q, self.q))

    def to_internal(self, value):
        return math.log(max(value, 1e-300))

    def from_internal(self, z):
        v = math.exp(z)
        return float(max(np.round(v / self.q) * self.q, self.q))

    def clip_internal(self, z):
        return float(min(max(z, self.low), self.high))


@dataclass
class RandInt(Expr):
    label: str
    upper: int

    def sample(self, rng):
        return int(rng.integers(0, self.upper))

    def from_internal(self, z):
        return int(min(max(round(z), 0), self.upper - 1))

    def clip_internal(self, z):
        return float(min(max(z, 0), self.upper - 1))


@dataclass
class Choice(Expr):
    label: str
    options: Sequence[Any]

    def sample(self, rng):
        return self.options[int(rng.integers(0, len(self.options)))]

    def to_internal(self, value):
        # store the option index
        for i, o in enumerate(self.options):
            if o == value or o is value:
                return float(i)
        raise ValueError(f"{value!r} not an option of {self.label}")

    def from_internal(self, z):
        return self.options[int(min(max(round(z), 0), len(self.options) - 1))]

    def clip_internal(self, z):
        return float(min(max(z, 0), len(self.options) - 1))


@dataclass
class IntCast(Expr):
    """``scope.int(expr)`` — cast the inner expression's value to int."""
    inner: Expr

    @property
    def label(self):
        return self.inner.label

    def sample(self, rng):
        return int(self.inner.sample(rng))

    def to_internal(self, value):
        return self.inner.to_internal(value)

    def from_internal(self, z):
        return int(self.inner.from_internal(z))

    def clip_internal(self, z):
        return self.inner.clip_internal(z)


class _HP:
    """The ``hp`` namespace (hyperopt-compatible factories)."""

    @staticmethod
    def uniform(label, low, high):
        return Uniform(label, low, high)

    @staticmethod
    def quniform(label, low, high, q):
        return QUniform(label, low, high, q)

    @staticmethod
    def loguniform(label, low, high):
        return LogUniform(label, low, high)

    @staticmethod
    def qloguniform(label, low, high, q):
        return QLogUniform(label, low, high, q)

    @staticmethod
    def normal(label, mu, sigma):
        return Normal(label, mu, sigma)

    @staticmethod
    def lognormal(label, mu, sigma):
        return LogNormal(label, mu, sigma)

    @staticmethod
    def randint(label, upper):
        return RandInt(label, upper)

    @staticmethod
    def choice(label, options):
        return Choice(label, list(options))


hp = _HP()


class _Scope:
    @staticmethod
    def int(expr: Expr) -> IntCast:
        return IntCast(expr)


scope = _Scope()


def flatten_space(space) -> Dict[str, Expr]:
    """A space is a dict of label → Expr (possibly nested dicts)."""
    out: Dict[str, Expr] = {}

    def _walk(node, prefix=""):
        if isinstance(node, Expr):
            out[node.label] = node
        elif isinstance(node, dict):
            for v in node.values():
                _walk(v)
        elif isinstance(node, (list, tuple)):
            for v in node:
                _walk(v)
        else:
            raise TypeError(f"unsupported space node: {node!r}")

    _walk(space)
    return out


def space_eval(space, best: Dict[str, Any]):
    """hyperopt-compat: substitute ``fmin``'s best dict into the space.

    ``fmin`` returns option *indices* for ``hp.choice`` parameters (as
    hyperopt does); this maps them back to the option values and rebuilds
    the user's space structure.
    """
    values: Dict[str, Any] = {}
    for lbl, expr in flatten_space(space).items():
        v = best[lbl]
        base = expr.inner if isinstance(expr, IntCast) else expr
        if isinstance(base, Choice):
            v = base.options[int(v)]
        values[lbl] = v
    return bind_params(space, values)


def bind_params(space, values: Dict[str, Any]):
    """Rebuild the user's space structure with sampled values in place."""
    if isinstance(space, Expr):
        return values[space.label]
    if isinstance(space, dict):
        return {k: bind_params(v, values) for k, v in space.items()}
    if isinstance(space, (list, tuple)):
        return type(space)(bind_params(v, values) for v in space)
    raise TypeError(f"unsupported space node: {space!r}")
