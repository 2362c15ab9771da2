"""skellysim_amd — MI355X-native engine for SkellySim's hydrodynamic hot path.

Scope (BASELINE.json north_star / SURVEY.md §8): the Stokeslet / stresslet /
regularized-Stokeslet / rotlet N-body velocity evaluations behind SkellySim's
kernels::Evaluator seam (reference include/kernels.hpp:14-15), rebuilt from
scratch as hand-written HIP/CDNA4 (gfx950) fp64 kernels behind the reference's
own evaluator API, target-sharded across GPUs with RCCL over xGMI.

This package is the PRODUCT path: it requires the in-tree HIP extension
(libskellyhip.so) and a GPU to compute. It never falls back to CPU — the CPU
restatement under oracle/ is test infrastructure only.
"""

from .evaluator import (  # noqa: F401
    Evaluator,
    set_evaluator,
    stokeslet_direct_gpu,
    stresslet_direct_gpu,
    oseen_contract_direct_gpu,
    rotlet_gpu,
    stresslet_times_normal_times_density,
    stokeslet_device,
    stresslet_device,
    oseen_contract_device,
    rotlet_device,
    stresslet_normal_density_device,
    stresslet_times_normal_device,
    oseen_tensor_batched_device,
)
from .sharded import ShardedPairEvaluator, shard_sizes, allgather_rows  # noqa: F401

__version__ = "0.1.0"
