"""FedOpt strategy tests vs NumPy oracles of the reference update rules
(fedavg_eff.py:318-324, fednestorov.py:323-331,
fedmom.py:263-278, fedadam.py:295-314, fedyogi.py:299-320)."""

import numpy as np
import pytest
import torch

from photon_amd.fed.flat import FlatParams
from photon_amd.fed.strategies import dispatch_strategy
from photon_amd.models.mpt import MPTCausalLM, MPTConfig


@pytest.fixture
def layout():
    torch.manual_seed(3)
    m = MPTCausalLM(MPTConfig(d_model=16, n_heads=2, n_layers=1, max_seq_len=16,
                              vocab_size=32, attn_impl="torch"))
    return FlatParams(m)


def run_rounds(strategy, layout, rounds=3, seed=5):
    rng = np.random.default_rng(seed)
    x0 = rng.normal(size=layout.total).astype(np.float32)
    strategy.initialize(torch.from_numpy(x0.copy()))
    avgs = [rng.normal(size=layout.total).astype(np.float32) for _ in range(rounds)]
    for r, avg in enumerate(avgs, start=1):
        strategy.update(torch.from_numpy(avg.copy()), r, n_clients=4)
    return x0, avgs, strategy.params.numpy()


def test_fedavg_oracle(layout):
    s = dispatch_strategy("FEDAVG", layout, {"server_learning_rate": 0.5})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    for avg in avgs:
        g = x - avg
        x = x - 0.5 * g
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_nesterov_oracle(layout):
    slr, mu = 0.7, 0.7
    s = dispatch_strategy("NESTOROV", layout,
                          {"server_learning_rate": slr, "server_momentum": mu})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    m = np.zeros_like(x)
    for avg in avgs:
        g = x - avg
        m = mu * m + g
        g = g + mu * m
        x = x - slr * g
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_fedmom_oracle(layout):
    slr, mu = 1.0, 0.9
    s = dispatch_strategy("MOM", layout,
                          {"server_learning_rate": slr, "server_momentum": mu})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    v = np.zeros_like(x)
    for avg in avgs:
        g = x - avg
        v_new = x - slr * g
        x = (1 + mu) * v_new - mu * v
        v = v_new
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_fedadam_oracle(layout):
    eta, b1, b2, tau = 0.1, 0.9, 0.99, 1e-9
    s = dispatch_strategy("FEDADAM", layout,
                          {"eta": eta, "beta_1": b1, "beta_2": b2, "tau": tau})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    m = np.zeros_like(x)
    v = np.zeros_like(x)
    for r, avg in enumerate(avgs, start=1):
        g = x - avg
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        x = x + eta * (m / (1 - b1**r)) / (np.sqrt(v / (1 - b2**r)) + tau)
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_fedyogi_oracle(layout):
    eta, b1, b2, tau = 0.01, 0.9, 0.99, 1e-3
    s = dispatch_strategy("FEDYOGI", layout,
                          {"eta": eta, "beta_1": b1, "beta_2": b2, "tau": tau})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    m = np.zeros_like(x)
    v = np.zeros_like(x)
    for r, avg in enumerate(avgs, start=1):
        g = x - avg
        m = b1 * m + (1 - b1) * g
        g2 = g * g
        v = v + (1 - b2) * g2 * np.sign(g2 - v)
        x = x + eta * (m / (1 - b1**r)) / (np.sqrt(v / (1 - b2**r)) + tau)
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_scaling_fn(layout):
    s = dispatch_strategy("FEDAVG", layout,
                          {"server_learning_rate": 1.0, "scaling_fn": "sqrt"})
    x0, avgs, out = run_rounds(s, layout)
    x = x0.copy()
    for avg in avgs:
        g = x - 2.0 * avg  # sqrt(4 clients) = 2
        x = x - g
    np.testing.assert_allclose(out, x, rtol=1e-4, atol=1e-5)


def test_state_roundtrip(layout):
    s = dispatch_strategy("NESTOROV", layout, {})
    run_rounds(s, layout)
    state = {k: v.clone() for k, v in s.state_tensors().items()}
    s2 = dispatch_strategy("NESTOROV", layout, {})
    s2.initialize(torch.zeros(layout.total))
    s2.load_state_tensors(state)
    assert torch.equal(s2.params, s.params)
    assert torch.equal(s2.m1, s.m1)


def test_metrics_present(layout):
    s = dispatch_strategy("NESTOROV", layout, {})
    s.initialize(torch.randn(layout.total))
    m = s.update(torch.randn(layout.total), 1, 4)
    assert "l2_norm_pseudo_gradient" in m
    assert "l2_norm_momentum_vector" in m
    assert len(m["layerwise_l2_norms_pseudo_gradient"]) == len(layout.names)


# ---------------------------------------------------------------------------
# Property-based fuzzing (hypothesis): the strategies must match their NumPy
# oracles for arbitrary hyperparameters and multi-round sequences.
# ---------------------------------------------------------------------------
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(
    slr=st.floats(0.05, 1.5),
    mu=st.floats(0.0, 0.99),
    rounds=st.integers(1, 4),
    seed=st.integers(0, 10_000),
)
def test_nesterov_property(layout, slr, mu, rounds, seed):
    rng = np.random.default_rng(seed)
    x0 = rng.standard_normal(layout.total).astype(np.float32)
    strat = dispatch_strategy("NESTOROV", layout,
                              {"server_learning_rate": slr, "server_momentum": mu})
    strat.initialize(torch.from_numpy(x0.copy()))
    x = x0.copy()
    m = np.zeros_like(x)
    for r in range(1, rounds + 1):
        avg = rng.standard_normal(layout.total).astype(np.float32)
        strat.update(torch.from_numpy(avg.copy()), r, 2)
        g = x - avg
        m = mu * m + g
        gp = g + mu * m
        x = x - slr * gp
    assert np.allclose(strat.params.numpy(), x, atol=1e-4), (
        np.abs(strat.params.numpy() - x).max()
    )


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(
    eta=st.floats(1e-3, 0.5),
    b1=st.floats(0.0, 0.95),
    b2=st.floats(0.5, 0.999),
    tau=st.floats(1e-6, 1e-2),
    rounds=st.integers(1, 4),
    seed=st.integers(0, 10_000),
)
def test_fedadam_property(layout, eta, b1, b2, tau, rounds, seed):
    rng = np.random.default_rng(seed)
    x0 = rng.standard_normal(layout.total).astype(np.float32)
    strat = dispatch_strategy("FEDADAM", layout, {
        "eta": eta, "beta_1": b1, "beta_2": b2, "tau": tau})
    strat.initialize(torch.from_numpy(x0.copy()))
    x = x0.copy()
    m = np.zeros_like(x)
    v = np.zeros_like(x)
    for r in range(1, rounds + 1):
        avg = rng.standard_normal(layout.total).astype(np.float32)
        strat.update(torch.from_numpy(avg.copy()), r, 2)
        g = x - avg
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        mh = m / (1 - b1**r)
        vh = v / (1 - b2**r)
        # NOTE the "+": the reference applies x + eta*m_hat/(sqrt(v_hat)+tau)
        # with pseudo-gradient x - avg (fedadam.py:291-316, verified against
        # the source) — parity means matching that exact rule.
        x = x + eta * mh / (np.sqrt(vh) + tau)
    assert np.allclose(strat.params.numpy(), x, atol=1e-4), (
        np.abs(strat.params.numpy() - x).max()
    )
