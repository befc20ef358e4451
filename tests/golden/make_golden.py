#!/usr/bin/env python3
"""Generate committed golden parity fixtures from the compiled reference.

Runs oracle/_ref/conflux_ref (the reference's own LU loop built in-container,
recipe oracle/ref_build/) on:
  * the reference's hard-coded known-answer matrices (lu_params.hpp:157-363)
    for every N where a power-of-two grid divides the matrix —
    N in {8, 16, 20, 32} (N=9/27 need Px=3; the reference's own butterfly
    posts unmatched sends for non-power-of-two Px, see oracle/lu_oracle.py),
  * seeded random inputs (oracle.gen_input) on the BASELINE grid shapes,
and stores {input, perm, C(global factored matrix, pivoted rows)} per config
into golden.npz.  Tests then pin the numpy oracle (and, transitively, the GPU
engine) against these WITHOUT needing MPI/MKL at test time.

Run in the build container (needs /root/reference + /opt/conda MPICH/MKL):
    python3 tests/golden/make_golden.py
"""
import os
import subprocess
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from oracle import Params, gen_matrix  # noqa: E402

REPO = os.path.join(os.path.dirname(__file__), "..", "..")
REF_BIN = os.path.join(REPO, "oracle", "_ref", "conflux_ref")

# (tag, N, v, Px, Py, Pz, use_kat)
CONFIGS = [
    # NOTE: the reference sizes candidatePivotBuff as Ml*(v+1)
    # (conflux_opt.hpp:447) but tournament_rounds factors 2v rows of it
    # (:291), so it requires Ml >= 2v, i.e. N >= 2*v*Px.  All configs below
    # respect that.
    ("kat8_v2_221", 8, 2, 2, 2, 1, True),
    ("kat8_v2_222", 8, 2, 2, 2, 2, True),
    ("kat16_v4_221", 16, 4, 2, 2, 1, True),
    ("kat16_v2_441", 16, 2, 4, 4, 1, True),
    ("kat16_v4_222", 16, 4, 2, 2, 2, True),
    ("kat32_v8_221", 32, 8, 2, 2, 1, True),
    ("kat32_v8_222", 32, 8, 2, 2, 2, True),
    ("kat32_v4_441", 32, 4, 4, 4, 1, True),
    ("rnd64_v8_221", 64, 8, 2, 2, 1, False),
    ("rnd128_v16_222", 128, 16, 2, 2, 2, False),
    ("rnd128_v8_442", 128, 8, 4, 4, 2, False),
    ("rnd256_v32_222", 256, 32, 2, 2, 2, False),
]


def run_ref(N, v, Px, Py, Pz, infile, pref):
    env = dict(os.environ, MKL_THREADING_LAYER="GNU", OMP_NUM_THREADS="1",
               LD_LIBRARY_PATH="/opt/conda/lib")
    subprocess.run(
        ["/opt/conda/bin/mpiexec", "-n", str(Px * Py * Pz), REF_BIN,
         str(N), str(v), str(Px), str(Py), str(Pz), infile, pref, "1"],
        env=env, check=True, capture_output=True, timeout=600)


def gather(pref, suffix, p):
    """Assemble the global matrix from per-rank tile-cyclic dumps
    (owner map: layout.cpp:95-123; lu_comm rank order is MPI_Cart row-major)."""
    Px, Py, Pz, v, N = p.Px, p.Py, p.Pz, p.v, p.N
    G = np.zeros((N, N))
    for rank in range(Px * Py * Pz):
        pi = rank // (Py * Pz)
        pj = (rank // Pz) % Py
        pk = rank % Pz
        if pk:
            continue
        loc = np.fromfile(f"{pref}.{suffix}.r{rank}").reshape(p.Ml, p.Nl)
        for lti in range(p.tA11x):
            for ltj in range(p.tA11y):
                gti, gtj = lti * Px + pi, ltj * Py + pj
                G[gti * v:(gti + 1) * v, gtj * v:(gtj + 1) * v] = \
                    loc[lti * v:(lti + 1) * v, ltj * v:(ltj + 1) * v]
    return G


def main():
    out = {}
    with tempfile.TemporaryDirectory() as td:
        for tag, N, v, Px, Py, Pz, kat in CONFIGS:
            p = Params(N, v, Px, Py, Pz)
            assert p.N == N, f"{tag}: grid pads N to {p.N}"
            pref = os.path.join(td, tag)
            if kat:
                infile = "-"   # reference's own InitMatrix KAT fill
            else:
                infile = os.path.join(td, tag + ".in")
                gen_matrix(N).tofile(infile)
            run_ref(N, v, Px, Py, Pz, infile, pref)
            A = gather(pref, "A", p)          # the input as the reference saw it
            C = gather(pref, "C", p)          # factored matrix, pivoted rows
            perm = np.fromfile(pref + ".perm", dtype=np.int32)
            out[f"{tag}/A"] = A
            out[f"{tag}/C"] = C
            out[f"{tag}/perm"] = perm
            out[f"{tag}/cfg"] = np.array([N, v, Px, Py, Pz], dtype=np.int32)
            print(f"{tag}: captured (N={N} grid={Px}x{Py}x{Pz})")
    dst = os.path.join(os.path.dirname(__file__), "golden.npz")
    np.savez_compressed(dst, **out)
    print(f"wrote {dst} ({os.path.getsize(dst)//1024} KiB)")


if __name__ == "__main__":
    main()
