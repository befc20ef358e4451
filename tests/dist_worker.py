"""One rank of a multi-process distributed test run (tests/test_dist_rccl.py).

Launched once per rank with LD_PRELOAD=tests/shimccl.so and SHIMCCL_DIR set,
so every rank's Engine runs the REAL distributed (!sim) code path of
libconflux_lu.so — grouped send/recv choreography, depth reduces, the
dual-comm lookahead, distributed validation — with the mock file transport
standing in for RCCL (all ranks share one GPU).  With SHIMCCL_DIR fixed,
ncclGetUniqueId is deterministic, so every rank derives the same uid locally
and no out-of-band exchange is needed.

usage: dist_worker.py N v Px Py Pz rank reps out.npz [set_matrix|nopiv]

With the optional 9th arg "set_matrix", the rank uploads its tile-cyclic
local slice through conflux_lu_set_matrix_local (the documented drop-in
data path) instead of the device-side generator — same matrix either way.
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
    mode = sys.argv[9] if len(sys.argv) > 9 else ""
    use_set = mode == "set_matrix"
    with Engine(N, v, Px, Py, Pz, rank=rank, world=P, uid=uid) as e:
        e.store_factors(True)
        if mode == "nopiv":
            e.set_pivoting(0)
        ms = 0.0
        for _ in range(reps):
            if mode == "nopiv":
                e.init_matrix_spd(42)  # diagonally dominant input
                ms = e.factor()
                continue
            if use_set:
                from oracle import gen_matrix
                pi, pj, pk = rank // (Py * Pz), (rank // Pz) % Py, rank % Pz
                loc = np.zeros((e.Ml, e.Nl))
                if pk == 0:
                    A = gen_matrix(N)
                    for lti in range(e.Ml // v):
                        for ltj in range(e.Nl // v):
                            gi, gj = lti * Px + pi, ltj * Py + pj
                            loc[lti*v:(lti+1)*v, ltj*v:(ltj+1)*v] = \
                                A[gi*v:(gi+1)*v, gj*v:(gj+1)*v]
                e.set_matrix_local(loc)
            else:
                e.init_matrix(42)
            ms = e.factor()
        resid = e.validate()  # distributed collective (rank 0 -> broadcast)
        F = e.get_F_local()
        perm = e.get_perm()
    np.savez(out, F=F, perm=perm, resid=resid, ms=ms)


if __name__ == "__main__":
    main()
