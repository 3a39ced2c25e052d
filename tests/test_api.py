"""PumiTally C++ facade tests (API parity with the reference 4-call flow)."""
import os

import numpy as np
import pytest

import pumiumtally_amd as pt


@pytest.fixture
def osh_mesh(tmp_path):
    m = pt.build_box(1, 1, 1)
    d = str(tmp_path / "mesh.osh")
    m.write_osh(d)
    return d


def test_facade_full_flow(osh_mesh, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("PUMITALLY_DEVICE", "cpu")
    n = 5
    tally = pt.PumiTally(osh_mesh, n)
    init = np.tile([0.1, 0.4, 0.5], (n, 1)).ravel()
    tally.copy_initial_position(init)
    dest = np.tile([1.2, 0.4, 0.5], (n, 1)).ravel()
    flying = np.ones(n, np.int8)
    weights = np.ones(n)
    tally.move_to_next_location(init, dest, flying, weights)
    # parity: the flying array is consumed and zeroed by the call
    # (reference PumiTallyImpl.cpp:169-172)
    assert (np.asarray(flying) == 0).all()
    tally.write_tally_results()
    assert os.path.exists("fluxresult.vtk")
    text = open("fluxresult.vtk").read()
    assert "SCALARS flux double" in text
    # normalized flux of element 4 = 0.5*5 / (1/6) = 15
    import re
    flux_vals = re.search(r"SCALARS flux double 1\nLOOKUP_TABLE default\n((?:[^\n]*\n){6})", text)
    vals = [float(v) for v in flux_vals.group(1).split()]
    assert abs(vals[2] - 0.3 * n * 6) < 1e-6
    assert abs(vals[3] - 0.1 * n * 6) < 1e-6
    assert abs(vals[4] - 0.5 * n * 6) < 1e-6


def test_facade_size_check(osh_mesh, monkeypatch):
    monkeypatch.setenv("PUMITALLY_DEVICE", "cpu")
    tally = pt.PumiTally(osh_mesh, 5)
    with pytest.raises(RuntimeError):
        tally.copy_initial_position(np.zeros(7))


def test_facade_output_env(osh_mesh, tmp_path, monkeypatch):
    monkeypatch.setenv("PUMITALLY_DEVICE", "cpu")
    out = str(tmp_path / "custom.vtk")
    monkeypatch.setenv("PUMITALLY_OUTPUT", out)
    n = 2
    tally = pt.PumiTally(osh_mesh, n)
    init = np.tile([0.1, 0.4, 0.5], (n, 1)).ravel()
    tally.copy_initial_position(init)
    tally.write_tally_results()
    assert os.path.exists(out)


def test_facade_tally_times(osh_mesh, monkeypatch):
    monkeypatch.setenv("PUMITALLY_DEVICE", "cpu")
    n = 3
    tally = pt.PumiTally(osh_mesh, n)
    init = np.tile([0.1, 0.4, 0.5], (n, 1)).ravel()
    tally.copy_initial_position(init)
    dest = np.tile([0.9, 0.4, 0.5], (n, 1)).ravel()
    tally.move_to_next_location(init, dest, np.ones(n, np.int8), np.ones(n))
    t = tally.tally_times()
    assert t["initialization_time"] > 0
    assert t["total_time_to_tally"] > 0
    assert t["vtk_file_write_time"] == 0.0


def test_quickstart_example_runs(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    import runpy
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    runpy.run_path(os.path.join(repo, "examples", "quickstart.py"),
                   run_name="__main__")
    assert os.path.exists("quickstart_flux.vtk")


def test_mesh_cli(tmp_path):
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, PYTHONPATH=repo)
    osh = str(tmp_path / "b.osh")
    r = subprocess.run([sys.executable, "-m", "pumiumtally_amd.mesh.cli",
                        "box", osh, "--cells", "3"], env=env,
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    r = subprocess.run([sys.executable, "-m", "pumiumtally_amd.mesh.cli",
                        "describe", osh], env=env, capture_output=True, text=True)
    assert r.returncode == 0 and "elements : 162" in r.stdout


def test_move_from_device_validates_groups_and_responses():
    """ptr() accepts uint16 groups / float64 responses interfaces and
    rejects mismatched dtypes; the CPU engine then refuses move_device
    (the GPU-only path), proving validation ran first."""
    import numpy as np
    import pytest
    import pumiumtally_amd as pt

    class Fake:
        def __init__(self, shape, typestr):
            self.__cuda_array_interface__ = {
                "shape": shape, "typestr": typestr, "strides": None,
                "data": (16, False), "version": 2,
            }

    m = pt.build_box(1, 1, 1)
    n = 4
    eng = pt.TallyEngine(m, n, device="cpu", nscores=2)
    dest = Fake((n * 3,), "<f8")
    fly = Fake((n,), "|i1")
    w = Fake((n,), "<f8")
    g = Fake((n,), "<u2")
    r = Fake((n * 2,), "<f8")
    with pytest.raises(RuntimeError, match="GPU engine"):
        eng.move_from_device(dest, fly, w, sync_torch=False, groups=g,
                             responses=r)
    with pytest.raises(TypeError):
        eng.move_from_device(dest, fly, w, sync_torch=False,
                             groups=Fake((n,), "<i2"))
    with pytest.raises(TypeError):
        eng.move_from_device(dest, fly, w, sync_torch=False,
                             responses=Fake((n,), "<f8"))  # wrong size


def test_new_api_size_validation():
    """Bindings reject mis-sized groups/responses/weights arrays loudly."""
    import numpy as np
    import pytest
    import pumiumtally_amd as pt
    from pumiumtally_amd import _core

    m = pt.build_box(2, 2, 2)
    n = 10
    eng = pt.TallyEngine(m, n, device="cpu", ngroups=2, nscores=2)
    o = np.full(n * 3, 0.5)
    fly = np.ones(n, np.int8)
    w = np.ones(n)
    eng.copy_initial_position(o)
    with pytest.raises(RuntimeError, match="groups size"):
        eng.move(o, o, fly, w, groups=np.zeros(n - 1, np.uint16))
    with pytest.raises(RuntimeError, match="responses size"):
        eng.move(o, o, fly, w, responses=np.zeros((n, 3)))
    with pytest.raises(RuntimeError, match="responses size"):
        eng.move_continue(o, fly, w, responses=np.zeros(n - 1))
    with pytest.raises(RuntimeError, match="groups size"):
        eng.walk_raw(o, o, np.zeros(n, np.int32), w,
                     groups=np.zeros(2 * n, np.uint16))
    with pytest.raises(RuntimeError, match="responses size"):
        eng.walk_raw(o, o, np.zeros(n, np.int32), w,
                     responses=np.zeros(n))
    with pytest.raises(RuntimeError, match="weights size"):
        _core.partition_morton(m, 2, np.zeros(m.nelems - 1))

    from pumiumtally_amd.parallel.partition import PartitionedTally
    ptal = PartitionedTally(m, device="cpu", ngroups=2, nscores=2)
    oo = np.full((n, 3), 0.5)
    with pytest.raises(ValueError, match="groups size"):
        ptal.run_segments(oo, oo, w, groups=np.zeros(n - 1, np.uint16))
    with pytest.raises(ValueError, match="responses size"):
        ptal.run_segments(oo, oo, w, responses=np.zeros((n - 1, 2)))
    with pytest.raises(ValueError, match="one entry per element"):
        ptal.repartition(np.zeros(3))
