"""Regression: bitwise-exact handoff resume in the partitioned engine.

The world-2 stateful soak (tools/part_world2_soak.py) found a tiny
elementwise divergence (~5e-6 absolute, conservation exact) at one
handoff in ~10^5: a resumed walk re-based at the cut-crossing point
computed its remaining crossings from perturbed endpoints, and a track
passing within fp noise of a face-edge junction attributed its final
sliver to the adjacent tet.  Handoff records now carry the walk's
t-parametrization (wrap-segment origin, progress t, exited-from
element; csrc/core/walk.h walk_segment doc), so the receiving rank
replays the sender's fp decisions exactly and partitioned flux matches
the replicated engine elementwise to atomic-reassociation noise.

This test replays the EXACT configuration that exposed the bug (seed 5,
20k particles, 100k tets, 15% origin resampling, divergence formerly at
step 18) at a 1e-12 gate -- three orders below the old failure, three
above fp noise.
"""
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_world2_handoff_is_bitwise():
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "part_world2_soak.py"),
         "--steps", "18", "--particles", "20000", "--mesh-tets", "100000",
         "--device", "cpu", "--tol", "1e-12"],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PART_WORLD2_SOAK_OK" in r.stdout


def test_world2_localization_matches_full_mesh_at_scale():
    """Second divergence source (2.1e-03 after 25 steps at 400k
    particles): a resampled origin ~7e-7 inside an element absent from
    the rank's ghost ring was loose-claimed into the adjacent ghost
    element, starting the walk one element off the oracle.  Submesh
    localization now claims strict hits only; loose hits resolve on the
    full mesh.  Replays the exact failing scale at 1e-12."""
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "part_world2_soak.py"),
         "--steps", "25", "--particles", "400000", "--mesh-tets", "100000",
         "--device", "cpu", "--full-size", "--tol", "1e-12"],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PART_WORLD2_SOAK_OK" in r.stdout


def test_world4_roundtrip_grouped_scored():
    """World-4 with groups, scores, escapes and a mid-run checkpoint
    roundtrip -- the full parity matrix in one configuration (also
    regression for the harness's own merge fill: max(0, x) corrupted
    escaped particles' boundary-clip positions)."""
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "part_world2_soak.py"),
         "--ranks", "4", "--seed", "202", "--steps", "20", "--particles",
         "200000", "--mesh-tets", "100000", "--device", "cpu", "--full-size",
         "--ngroups", "2", "--nscores", "2", "--escape-frac", "0.08",
         "--state-roundtrip-every", "7", "--tol", "1e-12"],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PART_WORLD2_SOAK_OK" in r.stdout
