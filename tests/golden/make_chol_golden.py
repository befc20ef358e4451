#!/usr/bin/env python3
"""Capture Cholesky parity fixtures from the compiled reference CONFCHOX
miniapp (oracle/_ref/conflux_chol_ref, DEBUG build: dumps its own generated
input matrix and the factored L — reference CholeskyIO.cpp:100-160,
Cholesky.cpp:155-158, :738-772).

The reference's input dump stores only each tile's LOWER half (its
generator fills tiles with cblas_dsyrk 'L'); the algorithm, like ours,
never reads the upper triangle, so fixtures keep the raw dump.

Run in the build container: python3 tests/golden/make_chol_golden.py
"""
import os
import subprocess
import sys
import tempfile

import numpy as np

REPO = os.path.join(os.path.dirname(__file__), "..", "..")
BIN = os.path.join(REPO, "oracle", "_ref", "conflux_chol_ref")

CONFIGS = [
    ("chol256_v64_221", 256, 64, 2, 2, 1),
    ("chol256_v64_222", 256, 64, 2, 2, 2),
    ("chol512_v64_221", 512, 64, 2, 2, 1),
]


def main():
    out = {}
    for tag, N, v, Px, Py, Pz in CONFIGS:
        with tempfile.TemporaryDirectory() as td:
            os.makedirs(os.path.join(td, "data"))
            env = dict(os.environ, MKL_THREADING_LAYER="GNU",
                       OMP_NUM_THREADS="1", LD_LIBRARY_PATH="/opt/conda/lib")
            subprocess.run(
                ["/opt/conda/bin/mpiexec", "-n", str(Px * Py * Pz), BIN,
                 "--dim", str(N), "--tile", str(v), "--grid",
                 f"{Px},{Py},{Pz}", "--run", "1"],
                cwd=td, env=env, check=True, capture_output=True, timeout=600)
            A = np.fromfile(os.path.join(td, "data", f"input_{N}.bin"))
            L = np.fromfile(os.path.join(td, "data", f"output_{N}.bin"))
            out[f"{tag}/A"] = A.reshape(N, N)
            out[f"{tag}/L"] = np.tril(L.reshape(N, N))
            out[f"{tag}/cfg"] = np.array([N, v, Px, Py, Pz], dtype=np.int32)
            print(f"{tag}: captured")
    dst = os.path.join(os.path.dirname(__file__), "chol_golden.npz")
    np.savez_compressed(dst, **out)
    print(f"wrote {dst} ({os.path.getsize(dst)//1024} KiB)")


if __name__ == "__main__":
    main()
