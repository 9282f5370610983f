"""Generate tests/golden/transpose_goldens.npz: oracle outputs for fixed
seeded inputs, committed as regression pins (the reference ships no golden
files — SURVEY.md §8(c)).  Inputs are reproduced at test time from the same
seeded generator (tests/util.py seeded_parents, seed 0xC0FFEE)."""

import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.join(HERE, ".."))
sys.path.insert(0, os.path.join(HERE, "..", "..", "oracle"))
sys.path.insert(0, os.path.join(HERE, "..", ".."))

import oracle as orc  # noqa: E402
from util import seeded_parents  # noqa: E402

ID3 = (0, 1, 2)
CASES = [
    ((16, 21, 41), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (), "float64"),
    ((16, 21, 41), (2, 3), (0, 2), (1, 2, 0), (0, 1), (2, 1, 0), (), "float64"),
    ((8, 9, 10), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (3,), "float64"),
    ((16, 21, 41), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (), "complex64"),
    ((16, 21, 41), (1, 1), (1, 2), ID3, (0, 2), ID3, (), "float64"),
]


def main():
    out = {"ncases": np.int64(len(CASES))}
    for c, (dims, pdims, di, pi, do, po, extra, dt) in enumerate(CASES):
        dtype = np.dtype(dt)
        _, srcs = seeded_parents(dims, pdims, di, pi, extra, dtype)
        outs = orc.transpose_oracle(srcs, dims, pdims, di, pi, do, po, extra)
        out[f"case{c}_meta"] = np.str_(dt)
        out[f"case{c}_dims"] = np.array(dims)
        out[f"case{c}_pdims"] = np.array(pdims)
        out[f"case{c}_di"] = np.array(di)
        out[f"case{c}_pi"] = np.array(pi)
        out[f"case{c}_do"] = np.array(do)
        out[f"case{c}_po"] = np.array(po)
        out[f"case{c}_extra"] = np.array(extra, dtype=np.int64)
        for r, o in enumerate(outs):
            out[f"case{c}_rank{r}"] = o
    path = os.path.join(HERE, "transpose_goldens.npz")
    np.savez_compressed(path, **out)
    print(f"wrote {path} ({os.path.getsize(path)} bytes, {len(CASES)} cases)")


if __name__ == "__main__":
    main()
