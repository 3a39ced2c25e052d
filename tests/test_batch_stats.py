"""Batch statistics (MC uncertainty accounting; beyond the reference)."""
import numpy as np
import pytest

import pumiumtally_amd as pt


def run_batches(device):
    m = pt.build_box(3, 3, 3)
    n = 50
    e = pt.TallyEngine(m, n, device=device)
    rng = np.random.default_rng(9)
    totals = []
    for b in range(4):
        o = rng.uniform(0.1, 0.9, size=(n, 3))
        d = rng.uniform(0.1, 0.9, size=(n, 3))
        w = rng.uniform(0.2, 1.0, n)
        e.copy_initial_position(o.ravel())
        e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
        totals.append(e.flux().copy())
        e.end_batch()
    e.synchronize()
    # per-batch flux was zeroed after each end_batch
    assert np.allclose(e.flux(), 0.0)
    mean, rel = e.batch_statistics()
    ref = np.stack(totals)
    assert np.allclose(mean, ref.mean(axis=0), atol=1e-12)
    sem = np.sqrt(np.maximum(ref.var(axis=0), 0) / 3)
    ref_rel = np.divide(sem, np.abs(ref.mean(axis=0)),
                        out=np.zeros_like(sem), where=ref.mean(axis=0) != 0)
    assert np.allclose(rel, ref_rel, atol=1e-10)
    return mean


def test_batch_stats_cpu():
    run_batches("cpu")


@pytest.mark.gpu
def test_batch_stats_gpu():
    mc = run_batches("cpu")
    mg = run_batches("cuda")
    assert np.abs(mc - mg).max() < 1e-10 * max(1.0, np.abs(mc).max())


def test_batch_stats_requires_batches():
    m = pt.build_box(1, 1, 1)
    e = pt.TallyEngine(m, 2, device="cpu")
    with pytest.raises(RuntimeError):
        e.batch_statistics()
