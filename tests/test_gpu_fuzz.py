"""In-suite slice of the hardware differential fuzz campaign.

Runs 150 randomized GPU-vs-CPU trials from tools/fuzz_gpu_differential
(random box mesh, vacuum/reflective/periodic-x BC, 1-3 groups, 1-3
scores, up to 4k segments; HIP flux elementwise vs the CPU engine,
stateful PartitionedEngine(cuda) vs plain, fp32 conservation, zero
lost).  The full campaign (20k+ trials, results in profiles/README.md)
runs from tools/; this keeps a regression slice in every GPU CI pass.
"""
import os
import sys

import numpy as np
import pytest

import pumiumtally_amd as pt

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tools"))

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not pt.have_gpu(), reason="needs a GPU")
def test_gpu_differential_fuzz_slice():
    from fuzz_gpu_differential import one_trial
    rng = np.random.default_rng(2026)
    for t in range(150):
        one_trial(rng, t)
