"""Property-based oracle checks (hypothesis): structural invariants that
must hold for ANY input, complementing the fixed-vector tests in
test_oracle.py. All CPU."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from oracle.moe_oracle import OracleConfig, gate_forward, moe_forward, route_tokens

pytestmark = pytest.mark.filterwarnings("ignore")


def _mk(S, H, E, seed):
    g = np.random.default_rng(seed)
    x = g.standard_normal((S, H), dtype=np.float32)
    gw = g.standard_normal((H * E,), dtype=np.float32)
    return x, gw


@settings(max_examples=30, deadline=None)
@given(
    S=st.integers(1, 64),
    H=st.sampled_from([8, 16, 32]),
    E=st.sampled_from([1, 2, 4, 8, 16]),
    k=st.integers(1, 8),
    seed=st.integers(0, 10_000),
)
def test_topk_selection_properties(S, H, E, k, seed):
    k = min(k, E)
    x, gw = _mk(S, H, E, seed)
    cfg = OracleConfig(num_experts=E, expert_top_k=k, element="fp32")
    gate_out, topk, mCw, tie = gate_forward(x, gw.reshape(E, H), cfg)
    # indices valid and distinct per token
    assert topk.shape == (S, k)
    assert ((topk >= 0) & (topk < E)).all()
    for t in range(S):
        assert len(set(topk[t].tolist())) == k
    # probabilities: rows sum to 1 over the E real columns, pads zero
    probs = gate_out[:, :E]
    np.testing.assert_allclose(probs.sum(1), 1.0, atol=1e-4)
    assert (gate_out[:, E:] == 0).all()
    # mCw equals the sum of the selected probabilities
    np.testing.assert_allclose(
        mCw, probs[np.arange(S)[:, None], topk].sum(1), rtol=1e-5, atol=1e-6)
    # the selected experts dominate: min selected prob >= max unselected
    for t in range(S):
        unsel = np.setdiff1d(np.arange(E), topk[t])
        if len(unsel):
            assert probs[t, topk[t]].min() >= probs[t, unsel].max() - 1e-7


@settings(max_examples=30, deadline=None)
@given(
    S=st.integers(1, 96),
    E=st.sampled_from([1, 2, 4, 8, 32]),
    k=st.integers(1, 4),
    cf=st.integers(1, 3),
    drop=st.integers(0, 1),
    seed=st.integers(0, 10_000),
)
def test_routing_conservation(S, E, k, cf, drop, seed):
    k = min(k, E)
    g = np.random.default_rng(seed)
    topk = np.stack([g.permutation(E)[:k] for _ in range(S)])
    cfg = OracleConfig(num_experts=E, expert_top_k=k, capacity_factor=cf,
                       drop_tokens=drop, element="fp32")
    lists, eC, kept = route_tokens(topk, cfg, S)
    # raw counts conserve every assignment
    assert eC.sum() == S * k
    base = -(-S // E) if drop else S
    EC = base * cf * k
    # kept lists are clipped to capacity and ordered subsets of tokens
    for e in range(E):
        assert len(lists[e]) == min(eC[e], EC)
        assert all(0 <= t < S for t in lists[e])
        assert sorted(set(lists[e])) == sorted(lists[e])  # no dup, ascending
    # kept mask agrees with the lists
    assert kept.shape == (S, k)
    total_kept = sum(len(lists[e]) for e in range(E))
    assert int(kept.sum()) == total_kept
    if not drop:
        assert kept.all()


@settings(max_examples=10, deadline=None)
@given(seed=st.integers(0, 10_000))
def test_moe_forward_token_permutation_equivariance(seed):
    """Permuting input tokens permutes the output identically (no-drop:
    routing has no capacity interaction between tokens)."""
    S, H, P, E, k = 24, 16, 16, 4, 2
    g = np.random.default_rng(seed)
    x = g.standard_normal((S, H), dtype=np.float32)
    gw = g.standard_normal((H * E,), dtype=np.float32)
    ew = g.standard_normal((E, 2, P, H), dtype=np.float32)
    cfg = OracleConfig(num_experts=E, expert_top_k=k, drop_tokens=0,
                      element="fp32")
    r = moe_forward(x, gw, ew, cfg)
    perm = g.permutation(S)
    rp = moe_forward(x[perm], gw, ew, cfg)
    np.testing.assert_allclose(rp["moe_out"], r["moe_out"][perm],
                               rtol=1e-5, atol=1e-5)
    assert np.array_equal(rp["topk_idx"], r["topk_idx"][perm])
