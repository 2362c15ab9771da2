"""GPU tests of the sharded evaluation path with real HIP compute (the gpurun
box has one GPU, so world logic beyond rank 0 is covered by the gloo CPU tests
in test_sharding.py; here the compute leg and target-subset parity run on
device)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def rel(a, b):
    return np.linalg.norm(a - b) / max(np.linalg.norm(b), 1e-300)


@pytest.fixture(scope="module")
def ska(hip_lib_path):
    import skellysim_amd
    return skellysim_amd


def test_sharded_evaluator_hip_compute_no_dist(ska, oracle_mod):
    """ShardedPairEvaluator outside a process group == plain evaluation."""
    from skellysim_amd.sharded import ShardedPairEvaluator
    rng = np.random.default_rng(3)
    n_src, n_trg = 4096, 2048
    dev = torch.device("cuda:0")
    r_src = torch.from_numpy(rng.uniform(-1, 1, (n_src, 3))).to(dev)
    f_src = torch.from_numpy(rng.uniform(-1, 1, (n_src, 3))).to(dev)
    r_trg = torch.from_numpy(rng.uniform(-1, 1, (n_trg, 3))).to(dev)
    ev = ShardedPairEvaluator()
    u = ev(r_src, f_src, r_trg, eta=1.3)
    torch.cuda.synchronize()
    ref = oracle_mod.stokeslet(r_src.cpu().numpy(), f_src.cpu().numpy(),
                               r_trg.cpu().numpy(), 1.3)
    assert rel(u.cpu().numpy(), ref) < 1e-10


def test_target_shard_reassembly_matches_full_eval(ska):
    """Evaluating target shards separately (what each GPU does) and
    reassembling equals the full evaluation bitwise — the sharded result is
    exactly the single-GPU result."""
    from skellysim_amd.sharded import shard_range
    rng = np.random.default_rng(9)
    n_src, n_trg, world = 3000, 1501, 4
    r_src = rng.uniform(-1, 1, (n_src, 3))
    f_src = rng.uniform(-1, 1, (n_src, 3))
    r_trg = rng.uniform(-1, 1, (n_trg, 3))
    u_full = ska.stokeslet_direct_gpu(r_src, None, r_trg, f_src, None, 1.0)
    parts = []
    for rank in range(world):
        a, b = shard_range(n_trg, world, rank)
        parts.append(ska.stokeslet_direct_gpu(r_src, None, r_trg[a:b], f_src, None, 1.0))
    assert np.array_equal(np.vstack(parts), u_full)


def test_subset_parity_at_2e5_sources(ska, oracle_mod):
    """Large-cloud parity on a target subset: 2e5 sources vs oracle on 512
    targets (keeps the CPU oracle leg to seconds)."""
    rng = np.random.default_rng(100)
    n_src = 200_000
    r_src = rng.uniform(-1, 1, (n_src, 3))
    f_src = rng.uniform(-1, 1, (n_src, 3))
    r_trg = rng.uniform(-1, 1, (512, 3))
    u = ska.stokeslet_direct_gpu(r_src, None, r_trg, f_src, None, 1.0)
    ref = oracle_mod.stokeslet(r_src, f_src, r_trg, 1.0)
    assert rel(u, ref) < 1e-10
    u9 = ska.stresslet_direct_gpu(None, r_src, r_trg, None,
                                  rng.uniform(-1, 1, (n_src, 9)), 1.0)
    assert np.all(np.isfinite(u9))
