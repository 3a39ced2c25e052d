"""Config-5 rehearsal (Python twin of cpp/mock_transport.cpp): mock
event-based transport replaying the OpenMC call pattern against the
4-call facade -- reincarnated, stopped and escaped slots mixed in every
MoveToNextLocation, as in the reference's event-based integration
(/root/reference/README.md:106-134)."""
import numpy as np
import pytest

import pumiumtally_amd as pt


class MockHost:
    """The host app's own bookkeeping (the OpenMC side of the fence)."""

    def __init__(self, n, seed=987654321):
        self.n = n
        self.rng = np.random.default_rng(seed)
        self.pos = np.zeros((n, 3))
        self.alive = np.ones(n, dtype=bool)
        self.wgt = np.ones(n)
        for i in range(n):
            self._sample(i)

    def _sample(self, i):
        self.pos[i] = 0.05 + 0.15 * self.rng.random(3)
        self.wgt[i] = 0.5 + self.rng.random()
        self.alive[i] = True

    def event_step(self):
        n = self.n
        origin = np.empty((n, 3))
        dest = np.empty((n, 3))
        flying = np.zeros(n, dtype=np.int8)
        weights = np.empty(n)
        for i in range(n):
            if not self.alive[i]:
                if self.rng.random() < 0.35:
                    self._sample(i)  # reincarnate: origin CHANGES
                else:
                    origin[i] = dest[i] = self.pos[i]
                    weights[i] = self.wgt[i]
                    continue
            origin[i] = self.pos[i]
            mu = 2.0 * self.rng.random() - 1.0
            phi = 2.0 * np.pi * self.rng.random()
            st = np.sqrt(1.0 - mu * mu)
            ln = 0.02 + 0.4 * self.rng.random()
            d = np.array([ln * st * np.cos(phi), ln * st * np.sin(phi),
                          ln * mu])
            dest[i] = self.pos[i] + d
            flying[i] = 1
            weights[i] = self.wgt[i]
            self.pos[i] = dest[i]
            if np.any(dest[i] < 0.0) or np.any(dest[i] > 1.0):
                self.alive[i] = False  # host geometry kill; engine clips
            else:
                self.wgt[i] *= 0.85
                if self.wgt[i] < 0.25 and self.rng.random() < 0.5:
                    self.alive[i] = False
        return origin, dest, flying, weights


def _drive_engine(mesh, n, steps, device, seed=11):
    host = MockHost(n, seed)
    eng = pt.TallyEngine(mesh, n, device=device)
    eng.copy_initial_position(host.pos.ravel().copy())
    for _ in range(steps):
        origin, dest, flying, weights = host.event_step()
        eng.move(origin.ravel(), dest.ravel(), flying, weights)
    eng.synchronize()
    return eng


def test_mock_transport_sequencing_cpu():
    """The CPU engine survives the full reincarnation/stop/escape mix and
    conserves: every element's tally is non-negative and the run is
    deterministic (same stream twice -> identical flux)."""
    mesh = pt.build_box(6, 6, 6)
    e1 = _drive_engine(mesh, 1500, 12, "cpu")
    e2 = _drive_engine(mesh, 1500, 12, "cpu")
    f1, f2 = e1.flux(), e2.flux()
    assert np.array_equal(f1, f2)
    assert (np.asarray(f1) >= 0).all()
    assert f1.sum() > 0
    assert e1.stats()["lost_particles"] == 0
    assert e1.stats()["relocated"] > 0  # reincarnations actually happened


@pytest.mark.gpu
def test_mock_transport_gpu_matches_cpu():
    """Config-5 rehearsal on hardware: identical event streams through
    the CPU oracle and the GPU engine; flux must agree elementwise."""
    mesh = pt.build_box(8, 8, 8)
    n, steps = 4000, 20
    cpu = _drive_engine(mesh, n, steps, "cpu")
    gpu = _drive_engine(mesh, n, steps, "cuda:0")
    fc, fg = np.asarray(cpu.flux()), np.asarray(gpu.flux())
    denom = np.where(np.abs(fc) > 1e-30, np.abs(fc), 1.0)
    rel = np.abs(fc - fg) / denom
    assert rel.max() < 1e-9, rel.max()
    assert gpu.stats()["lost_particles"] == 0
    # committed particle state agrees too (positions + escape flags)
    assert np.allclose(np.asarray(cpu.positions()).ravel(),
                       np.asarray(gpu.positions()).ravel(), atol=1e-12)
    assert np.array_equal(cpu.escaped(), gpu.escaped())
