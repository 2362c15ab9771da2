"""Full-BASELINE-size property tests (at the metric sizes, where
the O(N^2) oracle is impractical, parity is checked through size-independent
properties: linearity in strengths, 1/eta scaling, shard reassembly,
determinism)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ska(hip_lib_path):
    import skellysim_amd
    return skellysim_amd


def test_n1e6_properties(ska):
    """Three evaluations at the metric point N=1e6 (seed 100, the bench
    cloud): linearity + eta-scaling + determinism."""
    rng = np.random.default_rng(100)
    n = 1_000_000
    dev = torch.device("cuda:0")
    pts = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    f = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    g = torch.flip(f, dims=[0]).contiguous()

    u_f = ska.stokeslet_device(pts, f, pts, 1.0)
    u_g = ska.stokeslet_device(pts, g, pts, 1.0)
    u_sum = ska.stokeslet_device(pts, (f + g).contiguous(), pts, 1.0)
    torch.cuda.synchronize()
    rel = float(torch.norm(u_sum - (u_f + u_g)) / torch.norm(u_sum))
    assert rel < 1e-12, rel
    assert torch.isfinite(u_f).all()

    u_eta = ska.stokeslet_device(pts, f, pts, 4.0)
    torch.cuda.synchronize()
    rel = float(torch.norm(u_f - 4.0 * u_eta) / torch.norm(u_f))
    assert rel < 1e-13, rel

    u_f2 = ska.stokeslet_device(pts, f, pts, 1.0)
    torch.cuda.synchronize()
    assert torch.equal(u_f, u_f2)


def test_n1e6_stresslet_oseen_properties(ska):
    """Full-size properties for the other two hot kernels: linearity in the
    source strengths and eta scaling at N=1e6."""
    rng = np.random.default_rng(100)
    n = 1_000_000
    dev = torch.device("cuda:0")
    pts = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    f9 = torch.from_numpy(rng.uniform(-1, 1, (n, 9))).to(dev)
    g9 = torch.flip(f9, dims=[0]).contiguous()
    u_f = ska.stresslet_device(pts, f9, pts, 1.0)
    u_g = ska.stresslet_device(pts, g9, pts, 1.0)
    u_s = ska.stresslet_device(pts, (f9 + g9).contiguous(), pts, 1.0)
    torch.cuda.synchronize()
    assert float(torch.norm(u_s - (u_f + u_g)) / torch.norm(u_s)) < 1e-12
    u_eta = ska.stresslet_device(pts, f9, pts, 4.0)
    torch.cuda.synchronize()
    assert float(torch.norm(u_f - 4.0 * u_eta) / torch.norm(u_f)) < 1e-13

    rho = torch.from_numpy(rng.uniform(-1, 1, (n, 3))).to(dev)
    u1 = ska.oseen_contract_device(pts, pts, rho, 1.0)
    u2 = ska.oseen_contract_device(pts, pts, rho, 2.0)
    torch.cuda.synchronize()
    assert float(torch.norm(u1 - 2.0 * u2) / torch.norm(u1)) < 1e-13
    assert torch.isfinite(u1).all()

    w1 = ska.rotlet_device(pts, pts, rho, 1.0)
    w2 = ska.rotlet_device(pts, pts, (2.0 * rho).contiguous(), 4.0)
    torch.cuda.synchronize()
    assert float(torch.norm(w1 - 2.0 * w2) / torch.norm(w1)) < 1e-13
    assert torch.isfinite(w1).all()


def test_n1e6_subset_parity(ska, oracle_mod):
    """Direct oracle parity on a 256-target subset of the N=1e6 cloud
    (2.56e8 pairs on the host cores — seconds)."""
    rng = np.random.default_rng(100)
    n = 1_000_000
    pts = rng.uniform(-1, 1, (n, 3))
    f = rng.uniform(-1, 1, (n, 3))
    dev = torch.device("cuda:0")
    sub = pts[:256]
    u = ska.stokeslet_device(torch.from_numpy(pts).to(dev),
                             torch.from_numpy(f).to(dev),
                             torch.from_numpy(sub).to(dev), 1.0)
    torch.cuda.synchronize()
    ref = oracle_mod.stokeslet(pts, f, sub, 1.0)
    rel = np.linalg.norm(u.cpu().numpy() - ref) / np.linalg.norm(ref)
    assert rel < 1e-10, rel
