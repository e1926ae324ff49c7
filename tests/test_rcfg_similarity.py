"""RCFG + stochastic similarity filter unit tests (engine core, CPU)."""
import pytest
import torch

from ai_rtc_agent_amd.engine.rcfg import ResidualCFG
from ai_rtc_agent_amd.engine.similarity import StochasticSimilarityFilter


def test_rcfg_none_passthrough():
    r = ResidualCFG("none", 0.0)
    eps = torch.randn(4, 8, 8, 4)
    assert torch.equal(r.apply(eps, 1), eps)


def test_rcfg_inactive_guidance_passthrough():
    # guidance <= 1.0 means inactive even for self (matches reference default
    # guidance 0.0 / cfg self: no guidance math in the hot path)
    r = ResidualCFG("self", 0.0)
    eps = torch.randn(4, 8, 8, 4)
    r.reset(torch.randn(4, 8, 8, 4))
    assert torch.equal(r.apply(eps, 1), eps)


def test_rcfg_full():
    r = ResidualCFG("full", 2.0)
    eps_u = torch.randn(4, 8, 8, 4)
    eps_c = torch.randn(4, 8, 8, 4)
    out = r.apply(torch.cat([eps_u, eps_c]), 1)
    assert torch.allclose(out, eps_u + 2.0 * (eps_c - eps_u))


def test_rcfg_self_uses_and_shifts_stock():
    r = ResidualCFG("self", 1.5, delta=1.0)
    stock = torch.randn(4, 8, 8, 4)
    r.reset(stock)
    eps = torch.randn(4, 8, 8, 4)
    out = r.apply(eps, 1)
    assert torch.allclose(out, eps + 0.5 * (eps - stock))
    # stock shifted: stage i's eps is stage i+1's negative residual
    assert torch.allclose(r.stock_noise[1:], eps[:-1])


def test_rcfg_initialize_seeds_stock():
    r = ResidualCFG("initialize", 2.0)
    r.reset(torch.zeros(4, 8, 8, 4))
    seed = torch.randn(1, 8, 8, 4)
    eps_c = torch.randn(4, 8, 8, 4)
    out = r.apply(torch.cat([seed, eps_c]), 1)
    assert out.shape == (4, 8, 8, 4)


def test_rcfg_rejects_unknown():
    with pytest.raises(ValueError):
        ResidualCFG("bogus", 1.0)


def test_similarity_filter_skips_static_scene():
    g = torch.Generator().manual_seed(0)
    f = StochasticSimilarityFilter(threshold=0.98, max_skip_frame=10, generator=g)
    x = torch.randn(3, 64, 64)
    assert not f.should_skip(x)  # first frame never skips
    skips = sum(f.should_skip(x) for _ in range(50))
    assert skips > 30, "identical frames should mostly skip"


def test_similarity_filter_max_skip_cap():
    g = torch.Generator().manual_seed(0)
    f = StochasticSimilarityFilter(threshold=0.5, max_skip_frame=3, generator=g)
    x = torch.ones(3, 16, 16)
    f.should_skip(x)
    consec = 0
    longest = 0
    for _ in range(40):
        if f.should_skip(x):
            consec += 1
            longest = max(longest, consec)
        else:
            consec = 0
    assert longest <= 3


def test_similarity_filter_motion_resumes():
    g = torch.Generator().manual_seed(0)
    f = StochasticSimilarityFilter(threshold=0.98, generator=g)
    x = torch.randn(3, 64, 64)
    f.should_skip(x)
    assert not f.should_skip(torch.randn(3, 64, 64)), "different frame must run"
