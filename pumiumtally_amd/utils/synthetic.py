"""Synthetic straight-line particle histories for benchmarking.

There is no network access for datasets; BASELINE.json prescribes synthetic
straight-line histories with random source positions.  We generate, once,
two endpoint sets P0 -> P1 inside the box: random interior origins,
isotropic directions, segment lengths ~ mean_chord elements.  The bench
ping-pongs P0<->P1 so every step walks a full random segment set with zero
host-side work inside the timed region.
"""
from __future__ import annotations

import numpy as np


def _morton_keys(p, box, bits=10):
    """Interleaved-bit (Morton/Z-order) keys of points p in box."""
    scale = (1 << bits) - 1
    q = (p / np.asarray(box) * scale).astype(np.uint64)
    key = np.zeros(len(p), dtype=np.uint64)
    for b in range(bits):
        for axis in range(3):
            key |= ((q[:, axis] >> np.uint64(b)) & np.uint64(1)) << np.uint64(3 * b + axis)
    return key


def make_box_histories(box, n: int, mean_chord_elems: float, cells_per_axis: int,
                       seed: int = 0, pinned: bool = True, sort: bool = True,
                       source_frac: float = 1.0):
    """Returns (p0, p1, flying, weights) arrays for n particles in a box
    mesh of `cells_per_axis` cells per axis over extents `box` (3-tuple).

    mean_chord_elems: target mean number of element crossings per segment
    (an element is ~1/6 of a grid cell; a chord of k cells crosses ~2.2*k
    tets for this 6-tet cell cut, measured empirically).

    source_frac: origins are sampled inside the central cube covering this
    fraction of each axis.  1.0 = whole box (spread sources, the default
    steady-state load); small values (e.g. 0.02 = a couple of grid cells)
    put every particle through the same few elements -- the BASELINE
    config-4 atomic-contention stress, where thousands of lanes
    atomicAdd into the handful of tets around the source every step.
    """
    from .. import pinned_array

    rng = np.random.default_rng(seed)
    box = np.asarray(box, dtype=np.float64)
    cell = box / cells_per_axis
    # segment length targeting mean_chord_elems element crossings
    seg_len = float(mean_chord_elems / 2.2 * cell.mean())

    def alloc(shape, dtype="float64"):
        if pinned:
            return pinned_array(shape, dtype)
        return np.empty(shape, dtype=dtype)

    p0 = alloc((n, 3))
    p1 = alloc((n, 3))
    flying = alloc((n,), "int8")
    weights = alloc((n,), "float64")

    margin = 1e-6 * box
    if source_frac < 1.0:
        lo = box * (0.5 - source_frac / 2.0)
        hi = box * (0.5 + source_frac / 2.0)
        start = rng.uniform(np.maximum(lo, margin),
                            np.minimum(hi, box - margin), size=(n, 3))
    else:
        start = rng.uniform(margin, box - margin, size=(n, 3))
    if sort:
        # Spatial (Morton) ordering: adjacent particles walk adjacent mesh
        # regions, so each wave/XCD touches a compact working set (pairs
        # with the XCD-aware block remap in the walk kernel).  Event-based
        # MC codes sort their event queues the same way.
        start = start[np.argsort(_morton_keys(start, box))]
    p0[:] = start
    # isotropic directions
    u = rng.uniform(-1.0, 1.0, n)
    phi = rng.uniform(0.0, 2 * np.pi, n)
    s = np.sqrt(1.0 - u * u)
    d = np.stack([s * np.cos(phi), s * np.sin(phi), u], axis=1)
    end = p0 + seg_len * d
    # fold destinations back into the box (mirror reflection) so every
    # segment stays interior: steady-state walk load, no escapes.
    for k in range(3):
        end[:, k] = np.abs(end[:, k])
        over = end[:, k] > box[k]
        end[over, k] = 2 * box[k] - end[over, k]
        np.clip(end[:, k], margin[k], box[k] - margin[k], out=end[:, k])
    p1[:] = end
    flying[:] = 1
    weights[:] = rng.uniform(0.25, 1.0, n)
    return p0, p1, flying, weights
