"""world_size-2 gloo test of the distributed (N>1) choreography on CPU.

Runs the superstep loop SPMD over torch.distributed (gloo) on the 1x1x2
grid — the same communication pattern the engine's RCCL transport executes
for that grid (engine.cpp run_step, dist branches): C1/C7 depth reduces to
layer 0 with the fixed pk-ascending combine order, the C4 gpivots
broadcast, and the C8/C9 slab spreads from the layer-0 roots.  Compute per
rank uses the oracle's kernels (tests may use the oracle as checker).
Results must match the single-process oracle bit-for-bit on pivots.
"""
import os

import numpy as np
import pytest

from oracle import Params, gen_matrix, lu_oracle, residual_check
from oracle.lu_oracle import LUP


def _rank_main(rank, world, N, v, ret):
    import torch
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29511")
    dist.init_process_group("gloo", rank=rank, world_size=world)

    # grid 1x1xPz: pk == rank (engine rank map: (pi*Py + pj)*Pz + pk)
    p = Params(N, v, 1, 1, world)
    pk, layrK = rank, 0
    nlayr, Ml, Nl = p.nlayr, p.Ml, p.Nl

    A11 = gen_matrix(N) if pk == 0 else np.zeros((N, N))
    A10 = np.zeros((Ml, v))
    A01 = np.zeros((v, Nl))
    gri = np.arange(Ml, dtype=np.int64)
    fnp = 0
    pivotInds = np.full(N, -1, dtype=np.int64)
    Lg = np.zeros((N, N))
    U = np.zeros((N, N))

    def send(a, dst):
        dist.send(torch.from_numpy(np.ascontiguousarray(a)), dst=dst)

    def recv(shape, src):
        t = torch.zeros(*shape, dtype=torch.float64)
        dist.recv(t, src=src)
        return t.numpy()

    for k in range(p.Nt):
        off = loff = k * v
        nact = Ml - fnp
        # step 0: copy + depth reduce to layer 0 (C1), pk-ascending order
        A10[fnp:] = A11[fnp:, loff:loff + v]
        if pk != layrK:
            send(A10[fnp:], layrK)
        else:
            for src in range(1, world):
                A10[fnp:] += recv((nact, v), src)
        # step 1: LUP on layer 0 only; gpivots broadcast (C4)
        if pk == layrK:
            cand = np.zeros((max(2 * v, Ml), v + 1))
            cand[:nact, 0] = gri[fnp:].astype(np.float64)
            cand[:nact, 1:] = A10[fnp:]
            lu, perm = LUP(nact, v, cand)
            A00 = np.ascontiguousarray(lu[:v, :v])
            gpivots = cand[perm[:v], 0].astype(np.int64)
            for dst in range(1, world):
                send(gpivots.astype(np.float64), dst)
        else:
            gpivots = recv((v,), layrK).astype(np.int64)
        pivotInds[off:off + v] = gpivots
        # step 2: push pivots up (identical on every layer) + pack + reduce
        lrows = [int(np.where(gri == g)[0][0]) for g in gpivots]
        is_piv = np.zeros(Ml, bool)
        is_piv[lrows] = True
        early = [i for i in range(fnp, min(fnp + v, Ml)) if not is_piv[i]]
        late = [i for i in range(fnp + v, Ml) if is_piv[i]]

        def push(mat):
            tmp = mat[lrows].copy()
            mat[late] = mat[early]
            mat[fnp:fnp + v] = tmp

        push(A11)
        push(A10)
        g2 = gri.copy()
        t = g2[lrows].copy()
        g2[late] = g2[early]
        g2[fnp:fnp + v] = t
        gri = g2
        packed = A11[fnp:fnp + v, loff:].copy()
        if pk != layrK:
            send(packed, layrK)
        else:
            for src in range(1, world):
                packed += recv((v, Nl - loff), src)
            A01[:, :Nl - loff] = packed     # Px=1: order is identity
        fnp += v
        nact -= v
        # steps 4/5 on layer 0 + slab spreads (C8/C9)
        import scipy.linalg as la
        if pk == layrK:
            X = la.solve_triangular(np.triu(A00), A10[fnp:].T, trans="T",
                                    lower=False).T if nact else A10[fnp:]
            A10[fnp:] = X
            Lg[gri[fnp:], off:off + v] = X
            Y = la.solve_triangular(A00, A01[:, :Nl - loff], lower=True,
                                    unit_diagonal=True)
            A01[:, :Nl - loff] = Y
            U[off:off + v, off:] = Y[:, v:] if False else Y[:, :]
            U[off:off + v, off:off + v] = np.triu(A00)
            for i in range(v):
                Lg[gpivots[i], off:off + v] = np.tril(A00, -1)[i]
                Lg[gpivots[i], off + i] = 1.0
            # spread slabs: receiver (pk_rcv) gets cols/rows slab pk_rcv
            for dst in range(world):
                if dst == layrK:
                    continue
                send(X[:, dst * nlayr:(dst + 1) * nlayr], dst)
                send(Y[dst * nlayr:(dst + 1) * nlayr], dst)
            A10r = X[:, :nlayr]
            A01r = Y[:nlayr]
        else:
            A10r = recv((nact, nlayr), layrK)
            A01r = recv((nlayr, Nl - loff), layrK)
        # step 6: every layer multiplies its nlayr K-slice
        if nact:
            A11[fnp:, loff:] -= A10r @ A01r

    if rank == 0:
        F = U.copy()
        for r in range(N):
            F[r, :r] = Lg[pivotInds[r], :r]
        ret["perm"] = pivotInds
        ret["F"] = F
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gloo_world2_matches_oracle(tmp_path):
    import torch.multiprocessing as mp

    N, v = 64, 8
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29511"
    mgr = mp.Manager()
    ret = mgr.dict()
    mp.spawn(_rank_main, args=(2, N, v, ret), nprocs=2, join=True)
    A = gen_matrix(N)
    r = lu_oracle(A, Params(N, v, 1, 1, 2))
    assert np.array_equal(np.asarray(ret["perm"]), r["perm"]), \
        "gloo SPMD choreography must pick identical pivots"
    F = np.asarray(ret["F"])
    assert np.abs(F - r["F"]).max() < 1e-11
    assert residual_check(A, np.asarray(ret["perm"]), F) < 1e-14
