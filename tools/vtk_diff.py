#!/usr/bin/env python3
"""Diff two legacy-VTK tally files (the reference workflow's acceptance
check: fluxresult.vtk from this engine vs a reference run).

Usage: python tools/vtk_diff.py a.vtk b.vtk [--rtol 1e-9] [--atol 1e-10]
Exit code 0 if all cell-data fields match within tolerance.
"""
import argparse
import sys


def parse_vtk_cell_data(path):
    """Cell-data fields from a legacy VTK file, ASCII or BINARY."""
    import struct

    data = open(path, "rb").read()
    if b"\nBINARY\n" in data[:200]:
        fields = {}
        pos = data.index(b"CELL_DATA")
        ncells = int(data[pos:data.index(b"\n", pos)].split()[1])
        while True:
            pos = data.find(b"SCALARS", pos)
            if pos < 0:
                break
            name = data[pos:data.index(b"\n", pos)].split()[1].decode()
            pos = data.index(b"\n", data.index(b"LOOKUP_TABLE", pos)) + 1
            fields[name] = list(struct.unpack(f">{ncells}d",
                                              data[pos:pos + 8 * ncells]))
            pos += 8 * ncells
        return fields

    fields = {}
    lines = data.decode().splitlines()
    i = 0
    ncells = None
    while i < len(lines):
        line = lines[i]
        if line.startswith("CELL_DATA"):
            ncells = int(line.split()[1])
        elif line.startswith("SCALARS") and ncells is not None:
            name = line.split()[1]
            i += 1  # LOOKUP_TABLE
            vals = []
            while len(vals) < ncells:
                i += 1
                vals.extend(float(v) for v in lines[i].split())
            fields[name] = vals
        i += 1
    return fields


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("a")
    ap.add_argument("b")
    ap.add_argument("--rtol", type=float, default=1e-9)
    ap.add_argument("--atol", type=float, default=1e-10)
    args = ap.parse_args()
    fa, fb = parse_vtk_cell_data(args.a), parse_vtk_cell_data(args.b)
    if set(fa) != set(fb):
        print(f"field mismatch: {sorted(fa)} vs {sorted(fb)}")
        return 1
    worst = 0.0
    for name in fa:
        va, vb = fa[name], fb[name]
        if len(va) != len(vb):
            print(f"{name}: length {len(va)} vs {len(vb)}")
            return 1
        for x, y in zip(va, vb):
            err = abs(x - y) / (args.atol / args.rtol + max(abs(x), abs(y)))
            worst = max(worst, err)
            if abs(x - y) > args.atol + args.rtol * max(abs(x), abs(y)):
                print(f"{name}: {x} vs {y} differ beyond tolerance")
                return 1
        print(f"{name}: {len(va)} values match (worst rel err ~{worst:.2e})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
