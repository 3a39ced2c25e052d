"""Checkpoint/resume: a restored engine continues a batch identically."""
import numpy as np
import pytest

import pumiumtally_amd as pt


def make_history(rng, n, steps):
    segs = []
    cur = rng.uniform(0.05, 0.95, size=(n, 3))
    for _ in range(steps):
        nxt = np.clip(cur + rng.normal(0, 0.12, size=(n, 3)), 0.02, 0.98)
        segs.append((cur, nxt))
        cur = nxt
    return segs


def run(e, segs, w):
    n = len(w)
    for o, d in segs:
        e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)


def _roundtrip(device, tmp_path):
    m = pt.build_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(8)
    segs = make_history(rng, n, 6)
    w = rng.uniform(0.2, 1.0, n)

    # uninterrupted run
    ref = pt.TallyEngine(m, n, device=device)
    ref.copy_initial_position(segs[0][0].ravel())
    run(ref, segs, w)

    # checkpoint after 3 steps, resume in a NEW engine
    a = pt.TallyEngine(m, n, device=device)
    a.copy_initial_position(segs[0][0].ravel())
    run(a, segs[:3], w)
    ckpt = str(tmp_path / "state.npz")
    a.save_checkpoint(ckpt)
    del a

    b = pt.TallyEngine(m, n, device=device)
    b.load_checkpoint(ckpt)
    run(b, segs[3:], w)

    assert np.array_equal(ref.elem_ids(), b.elem_ids())
    assert np.allclose(ref.positions(), b.positions(), atol=0, rtol=0)
    assert np.abs(ref.flux() - b.flux()).max() < 1e-12


def test_checkpoint_roundtrip_cpu(tmp_path):
    _roundtrip("cpu", tmp_path)


@pytest.mark.gpu
def test_checkpoint_roundtrip_gpu(tmp_path):
    _roundtrip("cuda", tmp_path)


def test_checkpoint_shape_mismatch(tmp_path):
    m = pt.build_box(2, 2, 2)
    e = pt.TallyEngine(m, 10, device="cpu")
    p = str(tmp_path / "c.npz")
    e.save_checkpoint(p)
    e2 = pt.TallyEngine(m, 11, device="cpu")
    with pytest.raises(ValueError):
        e2.load_checkpoint(p)


def test_checkpoint_scored_grouped(tmp_path):
    """Checkpoint round-trip preserves the full (S,G,nelems) tally."""
    import numpy as np
    import pumiumtally_amd as pt

    m = pt.build_box(2, 2, 2)
    n, G, S = 40, 2, 2
    rng = np.random.default_rng(14)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = rng.uniform(0.1, 0.9, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, G, n).astype(np.uint16)
    r = rng.uniform(0.2, 2.0, size=(n, S))

    eng = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    f0 = eng.flux().copy()
    p0 = eng.positions().copy()
    path = str(tmp_path / "scored.npz")
    eng.save_checkpoint(path)

    # walk further, then restore: state must be exactly the checkpoint
    eng.move(d.ravel(), o.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    assert not np.allclose(eng.flux(), f0)
    eng.load_checkpoint(path)
    assert np.array_equal(eng.flux(), f0)
    assert np.array_equal(eng.positions(), p0)

    # resumed walk from the restored state matches a never-checkpointed run
    eng.move(d.ravel(), o.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    ref = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    ref.move(d.ravel(), o.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    assert np.allclose(eng.flux(), ref.flux(), atol=1e-14)
