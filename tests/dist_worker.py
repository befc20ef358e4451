"""One rank of a multi-process distributed test run (tests/test_dist_rccl.py).

Launched once per rank with LD_PRELOAD=tests/shimccl.so and SHIMCCL_DIR set,
so every rank's Engine runs the REAL distributed (!sim) code path of
libconflux_lu.so — grouped send/recv choreography, depth reduces, the
dual-comm lookahead, distributed validation — with the mock file transport
standing in for RCCL (all ranks share one GPU).  With SHIMCCL_DIR fixed,
ncclGetUniqueId is deterministic, so every rank derives the same uid locally
and no out-of-band exchange is needed.

usage: dist_worker.py N v Px Py Pz rank reps out.npz
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))


def main():
    import numpy as np

    N, v, Px, Py, Pz, rank, reps = map(int, sys.argv[1:8])
    out = sys.argv[8]
    from conflux_amd import Engine

    P = Px * Py * Pz
    uid = Engine.make_uid()
    with Engine(N, v, Px, Py, Pz, rank=rank, world=P, uid=uid) as e:
        e.store_factors(True)
        ms = 0.0
        for _ in range(reps):
            e.init_matrix(42)
            ms = e.factor()
        resid = e.validate()  # distributed collective (rank 0 -> broadcast)
        F = e.get_F_local()
        perm = e.get_perm()
    np.savez(out, F=F, perm=perm, resid=resid, ms=ms)


if __name__ == "__main__":
    main()
