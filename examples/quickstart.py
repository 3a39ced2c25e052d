"""Quickstart: build a mesh, tally a batch of particle tracks, write VTK.

Runs on CPU anywhere; add device="cuda:0" on an MI355X.
    python examples/quickstart.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402

# 1. a mesh: generated box here; pt.read_gmsh("reactor.msh") /
#    pt.read_mesh("mesh.osh") for real geometry
mesh = pt.build_box(10, 10, 10)
print(f"mesh: {mesh.nelems} tets")

# 2. an engine sized for the particle batch
n = 5000
eng = pt.TallyEngine(mesh, n, device="auto")

# 3. localize the sampled source positions (once per batch)
rng = np.random.default_rng(0)
pos = rng.uniform(0.05, 0.95, size=(n, 3))
eng.copy_initial_position(pos.ravel())

# 4. transport steps: each move tallies track_length * weight per element
weights = rng.uniform(0.5, 1.0, n)
alive = np.ones(n, np.int8)
for step in range(20):
    dest = np.clip(pos + rng.normal(0, 0.08, size=(n, 3)), 0.0, 1.0)
    eng.move(pos.ravel(), dest.ravel(), alive.copy(), weights)
    esc = eng.escaped().astype(bool)
    alive[esc] = 0                 # vacuum boundary: particle leaves
    pos = eng.positions()          # committed (possibly clipped) positions

# 5. results
flux = eng.flux()
print(f"total track length: {flux.sum():.3f}, alive {alive.sum()}/{n}")
eng.write_tally_results("quickstart_flux.vtk")
print("wrote quickstart_flux.vtk (cell data: flux, volume)")
