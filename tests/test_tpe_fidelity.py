"""TPE fidelity evidence (VERDICT r1 missing #5 / SURVEY §7 hard-part 4).

Asserts the adaptive-Parzen internals against hyperopt's published
algorithm step-by-step (hyperopt/tpe.py: ``default_gamma``,
``adaptive_parzen_normal``, ``linear_forgetting_weights``) and pins
seeded proposal traces for the three space types the reference uses
(``group_apply/02_...py:291-295`` scope.int(quniform);
``hyperopt/1. hyperopt.py:72`` lognormal;
``hyperopt/2...py:52`` uniform). hyperopt itself is not installable
here, so bit-for-bit draw parity is out of reach (documented divergence
list in tune/tpe.py's docstring); what IS checkable — and checked — is
formula-for-formula agreement of the posterior machinery plus
cross-process determinism of our own seeded traces.
"""
import json
import math
import os

import numpy as np
import pytest

from mi355x_scale.tune import fmin, hp, scope, tpe, Trials
from mi355x_scale.tune.tpe import (GAMMA, LINEAR_FORGETTING, PRIOR_WEIGHT,
                                   TPE, _adaptive_parzen,
                                   _forgetting_weights, _split)

FIXTURE = os.path.join(os.path.dirname(__file__), "fixtures",
                       "tpe_traces.json")


# ---------------------------------------------------------------- internals
def test_split_count_matches_hyperopt_default_gamma():
    """n_below = min(ceil(0.25·sqrt(n)), 25) — hyperopt default_gamma."""
    for n, want in [(10, 1), (16, 1), (17, 2), (64, 2), (100, 3),
                    (256, 4), (2500, 13), (10000, 25), (100000, 25)]:
        got = _split(np.zeros(n))
        assert got == want, (n, got, want)
    # degenerate guard: never the whole set
    assert _split(np.zeros(2)) == 1


def test_forgetting_weights_match_hyperopt_linear_forgetting():
    """m <= 25: all ones. m > 25: oldest m-25 ramp linspace(1/m, 1),
    newest 25 keep weight 1 — hyperopt linear_forgetting_weights."""
    assert (_forgetting_weights(10) == 1).all()
    assert (_forgetting_weights(25) == 1).all()
    w = _forgetting_weights(30)
    assert len(w) == 30
    assert (w[5:] == 1).all()
    np.testing.assert_allclose(w[:5], np.linspace(1 / 30, 1.0, num=5))


def test_adaptive_parzen_bandwidths_and_clamps():
    """Per-component sigma = max distance to adjacent sorted means with
    the prior inserted; clamps max=prior_sigma, min=prior_sigma/(1+m)
    (m<99) — hyperopt adaptive_parzen_normal."""
    prior_mu, prior_sigma = 0.0, 10.0
    mus = np.array([-4.0, 2.0, 5.0])
    w, means, sig = _adaptive_parzen(mus, prior_mu, prior_sigma)
    # means: observations in trial order + prior appended
    np.testing.assert_array_equal(means, [-4.0, 2.0, 5.0, 0.0])
    # sorted means: [-4, 0, 2, 5]
    # -4 is leftmost: left edge = -4 - prior_sigma -> dist 10; right 0 -> 4
    assert sig[0] == 10.0
    # 2: neighbors 0 and 5 -> max(2, 3) = 3
    assert sig[1] == 3.0
    # 5 is rightmost: left 2 -> 3; right edge 5 + prior_sigma -> 10
    assert sig[2] == 10.0
    w2, means2, sig2 = _adaptive_parzen(np.array([1.0]), 0.0, 10.0)
    # single obs at 1: sorted [0, 1]; left neighbor 0 -> 1, right edge 1+10
    assert sig2[0] == max(abs(1 - 0), abs((1 + 10) - 1))
    # clamp floor
    tight = _adaptive_parzen(np.array([1.0, 1.0, 1.0]), 0.0, 10.0)[2]
    assert tight[:3].min() >= 10.0 / min(100.0, 4.0)
    # clamp ceiling + prior keeps full width
    spread = _adaptive_parzen(np.array([-1e6, 1e6]), 0.0, 10.0)[2]
    assert spread.max() <= 10.0
    assert spread[-1] == 10.0


def test_adaptive_parzen_prior_weight_and_normalization():
    w, _, _ = _adaptive_parzen(np.arange(5.0), 0.0, 10.0)
    assert w.shape == (6,)
    np.testing.assert_allclose(w.sum(), 1.0)
    # all obs weight 1 (m<LF), prior weight PRIOR_WEIGHT -> uniform here
    np.testing.assert_allclose(w, np.full(6, 1 / 6))
    w30, _, _ = _adaptive_parzen(np.arange(30.0), 0.0, 100.0)
    # forgetting ramp downweights the oldest observations
    assert w30[0] < w30[-2]
    assert w30[-1] == pytest.approx(PRIOR_WEIGHT / (
        _forgetting_weights(30).sum() + PRIOR_WEIGHT))


def test_constants_match_hyperopt():
    assert GAMMA == 0.25
    assert LINEAR_FORGETTING == 25
    assert PRIOR_WEIGHT == 1.0
    # documented divergence: n_startup 10 (hyperopt 20) so the
    # reference's max_evals=10 inner searches engage the posterior
    assert TPE().n_startup == 10
    assert TPE().n_ei == 24


# ---------------------------------------------------------------- traces
def _reference_spaces():
    return {
        # W1 nested SARIMAX search (group_apply/02_...py:291-295)
        "sarimax_pdq": {
            "p": scope.int(hp.quniform("p", 0, 4, 1)),
            "d": scope.int(hp.quniform("d", 0, 2, 1)),
            "q": scope.int(hp.quniform("q", 0, 4, 1)),
        },
        # W2 SVC C (hyperopt/1. hyperopt.py:72)
        "svc_C": {"C": hp.lognormal("C", 0, 1)},
        # W2 LASSO alpha (hyperopt/2...py:52)
        "lasso_alpha": {"alpha": hp.uniform("alpha", 0.0, 10.0)},
    }


def _trace(space, losses_fn, n=18, seed=123):
    """Deterministic proposal trace: propose → synthetic loss → repeat."""
    algo = TPE()
    rng = np.random.default_rng(seed)
    hist = []
    trace = []
    for _ in range(n):
        params = algo.propose(space, hist, rng)
        loss = losses_fn(params)
        hist.append((params, loss))
        trace.append({"params": {k: (float(v) if isinstance(v, float)
                                     else v) for k, v in params.items()},
                      "loss": float(loss)})
    return trace


def _make_traces():
    spaces = _reference_spaces()
    return {
        "sarimax_pdq": _trace(
            spaces["sarimax_pdq"],
            lambda p: (p["p"] - 1) ** 2 + p["d"] + (p["q"] - 1) ** 2),
        "svc_C": _trace(spaces["svc_C"],
                        lambda p: (math.log(p["C"]) - 0.5) ** 2),
        "lasso_alpha": _trace(spaces["lasso_alpha"],
                              lambda p: abs(p["alpha"] - 3.0)),
    }


def test_seeded_traces_match_committed_fixture():
    """The committed fixture pins the full seeded proposal sequence for
    the three reference space types: any change to the Parzen machinery,
    the space internals, or the RNG threading shows up as a diff here
    (and the fixture regenerates identically across processes — the
    determinism hyperopt gets from rstate)."""
    got = _make_traces()
    if not os.path.exists(FIXTURE):  # first generation: write + verify
        os.makedirs(os.path.dirname(FIXTURE), exist_ok=True)
        with open(FIXTURE, "w") as f:
            json.dump(got, f, indent=1, sort_keys=True)
    with open(FIXTURE) as f:
        want = json.load(f)
    assert set(got) == set(want)
    for key in want:
        assert len(got[key]) == len(want[key])
        for g, w in zip(got[key], want[key]):
            assert g["params"].keys() == w["params"].keys()
            for k in g["params"]:
                assert g["params"][k] == pytest.approx(w["params"][k],
                                                       rel=1e-12), (key, k)


def test_trace_engages_posterior_after_startup():
    """Past n_startup the proposals must concentrate near the optimum —
    the behavioral check that the traces aren't pure random sampling."""
    with open(FIXTURE) as f:
        traces = json.load(f)
    alpha_tail = [t["params"]["alpha"] for t in traces["lasso_alpha"][12:]]
    alpha_head = [t["params"]["alpha"] for t in traces["lasso_alpha"][:10]]
    err_tail = np.mean([abs(a - 3.0) for a in alpha_tail])
    err_head = np.mean([abs(a - 3.0) for a in alpha_head])
    assert err_tail < err_head
