"""Binary Gmsh .msh ingestion (v2.2 and v4.1).

Round-1 hard-rejected binary .msh; real meshes at the 1M-10M-tet scale
are rarely exported as ASCII (VERDICT missing item 4).  The oracle is
the ASCII reader over the same mesh: identical vertex coordinates,
connectivity, volumes.
"""
import struct

import numpy as np
import pytest

import pumiumtally_amd as pt


def _box_arrays(cells=3):
    m = pt.build_box(cells, cells, cells)
    coords = np.asarray(m.coords).reshape(-1, 3)
    tets = np.asarray(m.tet2vert).reshape(-1, 4)
    return m, coords, tets


def _write_ascii_v2(path, coords, tets):
    with open(path, "w") as f:
        f.write("$MeshFormat\n2.2 0 8\n$EndMeshFormat\n")
        f.write(f"$Nodes\n{len(coords)}\n")
        for i, (x, y, z) in enumerate(coords):
            f.write(f"{i+1} {float(x)!r} {float(y)!r} {float(z)!r}\n")
        f.write("$EndNodes\n")
        f.write(f"$Elements\n{len(tets)}\n")
        for i, t in enumerate(tets):
            f.write(f"{i+1} 4 2 0 1 {t[0]+1} {t[1]+1} {t[2]+1} {t[3]+1}\n")
        f.write("$EndElements\n")


def _write_binary_v2(path, coords, tets):
    with open(path, "wb") as f:
        f.write(b"$MeshFormat\n2.2 1 8\n")
        f.write(struct.pack("<i", 1))
        f.write(b"\n$EndMeshFormat\n")
        f.write(b"$Nodes\n" + str(len(coords)).encode() + b"\n")
        for i, (x, y, z) in enumerate(coords):
            f.write(struct.pack("<iddd", i + 1, x, y, z))
        f.write(b"\n$EndNodes\n")
        f.write(b"$Elements\n" + str(len(tets) + 1).encode() + b"\n")
        # a non-tet block first (one triangle, type 2) to prove skipping
        f.write(struct.pack("<iii", 2, 1, 2))
        f.write(struct.pack("<iiiiii", 9000, 0, 1, 1, 2, 3))
        # the tet block, 2 tags each
        f.write(struct.pack("<iii", 4, len(tets), 2))
        for i, t in enumerate(tets):
            f.write(struct.pack("<iiiiiii", i + 1, 0, 1,
                                t[0] + 1, t[1] + 1, t[2] + 1, t[3] + 1))
        f.write(b"\n$EndElements\n")


def _write_binary_v4(path, coords, tets):
    with open(path, "wb") as f:
        f.write(b"$MeshFormat\n4.1 1 8\n")
        f.write(struct.pack("<i", 1))
        f.write(b"\n$EndMeshFormat\n")
        f.write(b"$Nodes\n")
        n = len(coords)
        # two entity blocks to prove block handling
        n0 = n // 2
        f.write(struct.pack("<QQQQ", 2, n, 1, n))
        for blk, (lo, hi) in enumerate(((0, n0), (n0, n))):
            f.write(struct.pack("<iiiQ", 3, blk + 1, 0, hi - lo))
            for i in range(lo, hi):
                f.write(struct.pack("<Q", i + 1))
            for i in range(lo, hi):
                f.write(struct.pack("<ddd", *coords[i]))
        f.write(b"\n$EndNodes\n")
        f.write(b"$Elements\n")
        ne = len(tets)
        f.write(struct.pack("<QQQQ", 2, ne + 1, 1, ne + 1))
        # triangle block (skipped)
        f.write(struct.pack("<iiiQ", 2, 1, 2, 1))
        f.write(struct.pack("<QQQQ", ne + 1, 1, 2, 3))
        # tet block
        f.write(struct.pack("<iiiQ", 3, 1, 4, ne))
        for i, t in enumerate(tets):
            f.write(struct.pack("<QQQQQ", i + 1,
                                t[0] + 1, t[1] + 1, t[2] + 1, t[3] + 1))
        f.write(b"\n$EndElements\n")


def _check(mesh_path, ref):
    m = pt.read_gmsh(str(mesh_path))
    assert m.nelems == ref.nelems and m.nverts == ref.nverts
    assert np.allclose(np.asarray(m.volumes), np.asarray(ref.volumes))
    assert np.asarray(m.coords).size == np.asarray(ref.coords).size
    # total volume identical (connectivity-level agreement)
    assert abs(np.asarray(m.volumes).sum() -
               np.asarray(ref.volumes).sum()) < 1e-12
    # walk equivalence: one segment tallies identically
    e1 = pt.TallyEngine(m, 1, device="cpu")
    e2 = pt.TallyEngine(ref, 1, device="cpu")
    o = np.array([0.11, 0.22, 0.33])
    d = np.array([0.77, 0.66, 0.55])
    for e in (e1, e2):
        e.copy_initial_position(o)
        e.move(o, d, np.ones(1, np.int8), np.ones(1))
    assert np.allclose(e1.flux(), e2.flux(), atol=1e-15)


def test_gmsh_binary_v2_matches_ascii(tmp_path):
    ref, coords, tets = _box_arrays()
    a = tmp_path / "a.msh"
    b = tmp_path / "b2.msh"
    _write_ascii_v2(a, coords, tets)
    _write_binary_v2(b, coords, tets)
    ra = pt.read_gmsh(str(a))
    _check(b, ra)


def test_gmsh_binary_v4_matches_ascii(tmp_path):
    ref, coords, tets = _box_arrays()
    a = tmp_path / "a.msh"
    b = tmp_path / "b4.msh"
    _write_ascii_v2(a, coords, tets)
    _write_binary_v4(b, coords, tets)
    ra = pt.read_gmsh(str(a))
    _check(b, ra)


def test_gmsh_binary_truncated_fails_loudly(tmp_path):
    ref, coords, tets = _box_arrays(2)
    b = tmp_path / "t.msh"
    _write_binary_v4(b, coords, tets)
    data = b.read_bytes()
    b.write_bytes(data[: len(data) // 2])
    with pytest.raises(RuntimeError):
        pt.read_gmsh(str(b))
