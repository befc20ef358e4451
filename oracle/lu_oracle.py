"""CPU restatement of the reference CONFLUX LU hot path.

All-ranks-in-one-process simulator of conflux::LU_rep<double>
(reference: src/conflux/lu/conflux_opt.hpp:344-1827) on the 3D process grid
of conflux::lu_params (lu_params.hpp:49-108).  Each function cites the
reference lines it follows.  This is TEST INFRASTRUCTURE (see
oracle/__init__.py header): the product never calls it.

Restrictions (same as the engine, documented in DESIGN.md):
  * Px == Py — the reference's A00 transpose-pair exchange
    (conflux_opt.hpp:818-850) only pairs up when k%Px == k%Py, and its own
    grid heuristic assumes P = Px*Px*Pz (lu_params.hpp:56).
  * Px is a power of two — for other Px the reference's tournament posts
    sends that no rank ever receives (conflux_opt.hpp:253-280 pairs rank a
    with butterfly_pair(a) even when butterfly_pair(butterfly_pair(a)) != a,
    e.g. Px=3, round 1, rank 1 -> 2 while rank 2 pairs with 0), i.e. it
    relies on MPI eager buffering of permanently-unmatched messages.
    BASELINE grids use Px in {1, 2}.
  * M == N (the miniapp always passes M = N, conflux_miniapp.cpp:81).
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field

import numpy as np
from scipy.linalg import lapack as _lapack
from scipy.linalg import solve_triangular as _solve_tri


def flipbit(n: int, k: int) -> int:
    # conflux_opt.cpp:55-57
    return n ^ (1 << k)


def g2lnoTile(grows, size, Px, v):
    """conflux_opt.cpp:74-98: global pivot rows -> {owner pi: global rows},
    {owner pi: positions in gpivots (pivot order)}."""
    lrows: dict[int, list[int]] = {}
    loffsets: dict[int, list[int]] = {}
    for i in range(size):
        g = int(grows[i])
        pOwn = (g // v) % Px
        lrows.setdefault(pOwn, []).append(g)
        loffsets.setdefault(pOwn, []).append(i)
    return lrows, loffsets


@dataclass
class Params:
    """Derived sizes of lu_params<T>::initialize (lu_params.hpp:49-82)."""
    N: int
    v: int
    Px: int
    Py: int
    Pz: int

    def __post_init__(self):
        assert self.Px == self.Py, "Px == Py only (see module doc)"
        assert self.Px & (self.Px - 1) == 0, "power-of-two Px only (see module doc)"
        # nlayr = ceil(v/Pz) (lu_params.hpp:73) makes the pk = Pz-1 slab of the
        # step-4/5 splits (conflux_opt.hpp:1389-1399) run past v when
        # v % Pz != 0 — the reference fudges this; we require it.
        assert self.v % self.Pz == 0, "v must be divisible by Pz"
        v, Px, Py, Pz = self.v, self.Px, self.Py, self.Pz
        ntx = (self.N + v * Px - 1) // (v * Px)          # lu_params.hpp:67
        nty = (self.N + v * Py - 1) // (v * Py)          # lu_params.hpp:68
        self.M = v * Px * ntx                            # lu_params.hpp:70
        self.N = v * Py * nty                            # lu_params.hpp:71
        self.nlayr = (v + Pz - 1) // Pz                  # lu_params.hpp:73
        self.Nt = (self.N + v - 1) // v                  # lu_params.hpp:75
        self.Mt = (self.M + v - 1) // v                  # lu_params.hpp:76
        self.tA11x = (self.Mt + Px - 1) // Px            # lu_params.hpp:78
        self.tA11y = (self.Nt + Py - 1) // Py            # lu_params.hpp:79
        self.Ml = self.tA11x * v                         # lu_params.hpp:81
        self.Nl = self.tA11y * v                         # lu_params.hpp:82
        self.P = Px * Py * Pz


def distribute(A: np.ndarray, p: Params) -> dict:
    """Tile-cyclic owner map (layout.cpp:95-123): global tile (gti, gtj) is
    owned by rank (gti%Px, gtj%Py, 0) at local tile (gti//Px, gtj//Py);
    layers pk > 0 start as zeros (lu_params.hpp:150-155)."""
    A11 = {}
    v = p.v
    for pi in range(p.Px):
        for pj in range(p.Py):
            loc = np.zeros((p.Ml, p.Nl))
            for lti in range(p.tA11x):
                gti = lti * p.Px + pi
                for ltj in range(p.tA11y):
                    gtj = ltj * p.Py + pj
                    loc[lti * v:(lti + 1) * v, ltj * v:(ltj + 1) * v] = \
                        A[gti * v:(gti + 1) * v, gtj * v:(gtj + 1) * v]
            for pk in range(p.Pz):
                A11[(pi, pj, pk)] = loc.copy() if pk == 0 else np.zeros((p.Ml, p.Nl))
    return A11


def LUP(n_rows: int, v: int, cand: np.ndarray):
    """conflux_opt.hpp:143-166: factor cand[:n_rows, 1:v+1] (col 0 holds the
    glued global-row ids and is skipped); returns (packed LU factors, perm).

    perm: identity over max(2v, n_rows) entries with LAPACK's ipiv swaps
    applied left-to-right — factored row i came from input row perm[i]."""
    perm = np.arange(max(2 * v, n_rows), dtype=np.int64)
    if n_rows == 0:
        return np.zeros((0, v)), perm
    a = np.array(cand[:n_rows, 1:v + 1], dtype=np.float64, order="F")
    lu, piv, info = _lapack.dgetrf(a)            # scipy piv is 0-based
    assert info >= 0, f"dgetrf illegal arg {info}"
    for i in range(min(v, n_rows)):
        j = int(piv[i])
        perm[i], perm[j] = perm[j], perm[i]
    return np.ascontiguousarray(lu), perm


def lu_oracle(A: np.ndarray, p: Params, collect_steps: bool = False):
    """Run the full LU_rep superstep loop (conflux_opt.hpp:344-1827).

    Returns dict with:
      perm    : int64[M] — pivotIndsBuff: row r of PA is row perm[r] of A
      F       : fp64[N,N] — factored matrix in pivoted row order:
                strict lower = L multipliers (unit diag implied), upper = U
      gpivots : list of per-step global pivot id arrays
      A00s / A10s / A01s : per-step panel dumps when collect_steps
    """
    N, v, Px, Py, Pz = p.N, p.v, p.Px, p.Py, p.Pz
    nlayr, Nt, Ml, Nl = p.nlayr, p.Nt, p.Ml, p.Nl
    layrK = 0                                    # conflux_opt.hpp:552

    A11 = distribute(A, p)
    A10 = {key: np.zeros((Ml, v)) for key in A11}
    A01 = {key: np.zeros((v, Nl)) for key in A11}
    A10Rcv = {key: np.zeros((Ml, nlayr)) for key in A11}
    A01Rcv = {key: np.zeros((nlayr, Nl)) for key in A11}

    # global row index per local row (conflux_opt.hpp:427-440); identical
    # across (pj, pk) for fixed pi, so tracked once per pi
    gri = {pi: np.array([(i // v * Px + pi) * v + i % v for i in range(Ml)],
                        dtype=np.int64) for pi in range(Px)}
    fnp = {pi: 0 for pi in range(Px)}            # first_non_pivot_row
    nact = {pi: Ml for pi in range(Px)}          # n_local_active_rows

    pivotInds = np.full(p.M, -1, dtype=np.int64)
    Lg = np.zeros((N, N))                        # L rows keyed by GLOBAL row id
    U = np.zeros((N, N))                         # U rows in pivot order
    out = {"gpivots": [], "A00s": [], "A10s": [], "A01s": []}

    n_rounds = int(math.ceil(math.log2(Px))) if Px > 1 else 0

    for k in range(Nt):
        off = k * v
        loff = (k // Py) * v                     # conflux_opt.hpp:549
        kcol = k % Py
        krow = k % Px

        # ---- step 0: reduce first active tile column -> A10 on layer layrK
        # (conflux_opt.hpp:618-646)
        for pi in range(Px):
            f, n = fnp[pi], nact[pi]
            for pk in range(Pz):
                A10[(pi, kcol, pk)][f:f + n] = \
                    A11[(pi, kcol, pk)][f:f + n, loff:loff + v]
            A10[(pi, kcol, layrK)][f:f + n] = \
                sum(A10[(pi, kcol, pk)][f:f + n] for pk in range(Pz))

        # ---- step 1: tournament pivoting (conflux_opt.hpp:689-850)
        cand = {}
        A00 = {}
        for pi in range(Px):
            f, n = fnp[pi], nact[pi]
            c = np.zeros((max(2 * v, Ml), v + 1))
            c[:n, 0] = gri[pi][f:f + n].astype(np.float64)   # prepend_column
            c[:n, 1:] = A10[(pi, kcol, layrK)][f:f + n]
            lu, perm = LUP(n, v, c)                          # conflux_opt.hpp:727
            # winners = rows perm[0..v) incl. the id column; zero-padded rows
            # stand in when n < v (step0_padding, conflux_opt.hpp:604-614)
            winners = c[perm[:v], :].copy()
            # placement (conflux_opt.hpp:741-751): my winners sit in the
            # bottom half iff my round-0 partner has the lower coordinate
            buf = np.zeros((2 * v, v + 1))
            if Px > 1 and flipbit(pi, 0) < pi:
                buf[v:] = winners
            else:
                buf[:v] = winners
            cand[pi] = buf
            if n_rounds == 0:
                # Px == 1: tournament_rounds never runs (numRounds = 0,
                # conflux_opt.hpp:778) and the reference writes A00Buff only
                # inside the final round (conflux_opt.hpp:308-310) — a latent
                # gap for 1-wide grids.  Intent (verified against the
                # algorithm): A00 = top v x v of the factors.
                A00[pi] = lu[:v, :v].copy()
                cand[pi][:v] = winners

        # tournament rounds (conflux_opt.hpp:220-336).  For power-of-two Px
        # the sendrecv exchange reduces to: both members of pair (lo, hi)
        # end up with [lo's winner half ; hi's winner half] — the lower
        # coordinate's candidates always on top (conflux_opt.hpp:717-719).
        for r in range(n_rounds):
            merged = {}
            for pi in range(Px):
                src = flipbit(pi, r)
                lo, hi = min(pi, src), max(pi, src)
                m = np.zeros((2 * v, v + 1))
                m[:v] = cand[lo][:v]
                m[v:] = cand[hi][v:]
                merged[pi] = m
            for pi in range(Px):
                c = merged[pi]
                lu, perm = LUP(2 * v, v, c)
                winners = c[perm[:v], :].copy()
                buf = np.zeros((2 * v, v + 1))
                if r == n_rounds - 1:
                    buf[:v] = winners                        # conflux_opt.hpp:294-299
                    A00[pi] = lu[:v, :v].copy()              # conflux_opt.hpp:308-310
                else:
                    if flipbit(pi, r + 1) < pi:              # conflux_opt.hpp:312-323
                        buf[v:] = winners
                    else:
                        buf[:v] = winners
                cand[pi] = buf

        min_perm = min(N - k * v, v)                         # conflux_opt.hpp:692
        gpivots = cand[0][:min_perm, 0].astype(np.int64)     # conflux_opt.hpp:810-816
        A00blk = A00[0]
        for pi in range(1, Px):
            # all participants hold identical winners/A00 (same getrf inputs)
            assert np.array_equal(cand[pi][:min_perm, 0].astype(np.int64), gpivots)
            assert np.array_equal(A00[pi], A00blk)
        # A00 transpose-pair exchange (conflux_opt.hpp:818-850) then lands the
        # same A00blk on row pi == k%Px; gpivots broadcast (:871-873).

        pivotInds[off:off + min_perm] = gpivots              # conflux_opt.hpp:910
        out["gpivots"].append(gpivots.copy())
        if collect_steps:
            out["A00s"].append(A00blk.copy())

        lpivots, loffsets = g2lnoTile(gpivots, min_perm, Px, v)

        # ---- step 2: push pivot rows up + pack + depth-reduce
        # (conflux_opt.hpp:1020-1174, push: :176-218 + conflux_opt.cpp:100-148)
        reduced = {}   # (pi, pj) -> summed pivot rows, cols loff:
        order_by_pi = {}
        for pi in range(Px):
            rows_g = lpivots.get(pi, [])
            order_by_pi[pi] = loffsets.get(pi, [])
            cnt = len(rows_g)
            f = fnp[pi]
            igri = {int(g): i for i, g in enumerate(gri[pi])}
            lrows = [igri[g] for g in rows_g]
            is_piv = np.zeros(Ml, dtype=bool)
            is_piv[lrows] = True
            early_np = [i for i in range(f, min(f + cnt, Ml)) if not is_piv[i]]
            late_p = [i for i in range(f + cnt, Ml) if is_piv[i]]
            assert len(early_np) == len(late_p)

            def push(mat):
                tmp = mat[lrows].copy()
                mat[late_p] = mat[early_np]
                mat[f:f + cnt] = tmp

            for pj in range(Py):
                for pk in range(Pz):
                    push(A11[(pi, pj, pk)])
                    push(A10[(pi, pj, pk)])
            push(gri[pi])
            fnp[pi] += cnt
            nact[pi] -= cnt
            for pj in range(Py):
                reduced[(pi, pj)] = sum(
                    A11[(pi, pj, pk)][f:f + cnt, loff:] for pk in range(Pz))

        # ---- step 3: route pivot rows to row k%Px, pivot-ordered into A01
        # (conflux_opt.hpp:1191-1259, 1454-1513)
        for pj in range(Py):
            dst = A01[(krow, pj, layrK)]
            for pi in range(Px):
                order = order_by_pi[pi]
                acc = reduced[(pi, pj)]
                for i, o in enumerate(order):
                    dst[o, :Nl - loff] = acc[i]

        # ---- step 4: A10 <- A10 * U(A00)^-1 (cblas_dtrsm
        # Right/Upper/NoTrans/NonUnit, conflux_opt.hpp:1347-1358), slab-split
        # into Pz chunks of nlayr cols, spread over (pj, pk) (:1389-1434)
        Ublk = np.triu(A00blk)
        for pi in range(Px):
            f, n = fnp[pi], nact[pi]
            X0 = A10[(pi, kcol, layrK)][f:f + n]
            X = _solve_tri(Ublk, X0.T, trans="T", lower=False).T if n else X0[:0]
            A10[(pi, kcol, layrK)][f:f + n] = X
            Lg[gri[pi][f:f + n], off:off + v] = X
            if collect_steps:
                out["A10s"].append((k, pi, gri[pi][f:f + n].copy(), X.copy()))
            for pj in range(Py):
                for pk in range(Pz):
                    A10Rcv[(pi, pj, pk)][:n] = X[:, pk * nlayr:(pk + 1) * nlayr]

        # ---- step 5: A01 <- L(A00)^-1 * A01 (cblas_dtrsm
        # Left/Lower/NoTrans/Unit, conflux_opt.hpp:1539-1551), slab-split into
        # Pz chunks of nlayr rows, spread over (pi, pk) (:1568-1592)
        for pj in range(Py):
            Y0 = A01[(krow, pj, layrK)][:, :Nl - loff]
            Y = _solve_tri(A00blk, Y0, lower=True, unit_diagonal=True)
            A01[(krow, pj, layrK)][:, :Nl - loff] = Y
            if collect_steps:
                out["A01s"].append((k, pj, Y.copy()))
            # U rows (pivot order); columns left of the diagonal block carry
            # stale already-factored data the reference never reads — skip
            lcs = np.arange(loff, Nl)
            gcs = (lcs // v * Py + pj) * v + lcs % v
            sel = gcs >= off
            U[np.ix_(np.arange(off, off + min_perm), gcs[sel])] = Y[:min_perm, sel]
            for pi in range(Px):
                for pk in range(Pz):
                    A01Rcv[(pi, pj, pk)][:, :Nl - loff] = Y[pk * nlayr:(pk + 1) * nlayr]

        # diagonal block from A00 (reference validation convention,
        # conflux_opt.hpp:1743-1752)
        U[off:off + v, off:off + v] = np.triu(A00blk)
        tri = np.tril(A00blk, -1)
        for i in range(min_perm):
            Lg[gpivots[i], off:off + v] = tri[i]
            Lg[gpivots[i], off + i] = 1.0

        # ---- step 6: trailing update (conflux_opt.hpp:1628-1633)
        for pi in range(Px):
            f, n = fnp[pi], nact[pi]
            if n == 0:
                continue
            for pj in range(Py):
                for pk in range(Pz):
                    A11[(pi, pj, pk)][f:f + n, loff:] -= \
                        A10Rcv[(pi, pj, pk)][:n] @ A01Rcv[(pi, pj, pk)][:, :Nl - loff]

    # assemble F in pivoted row order: row r of PA = row perm[r] of A
    F = U.copy()
    for r in range(N):
        F[r, :r] = Lg[pivotInds[r], :r]
    out.update(perm=pivotInds, F=F, Lg=Lg, U=U)
    return out


def residual_check(A: np.ndarray, perm: np.ndarray, F: np.ndarray) -> float:
    """|| P A - L U ||_F / ||A||_F with F in pivoted row order."""
    N = A.shape[0]
    L = np.tril(F, -1) + np.eye(N)
    Uu = np.triu(F)
    PA = A[perm[:N]]
    return float(np.linalg.norm(PA - L @ Uu) / np.linalg.norm(A))


def lu_nopivot(A: np.ndarray, v: int) -> np.ndarray:
    """Blocked LU WITHOUT pivoting — the Python prototype's EmptyPivot
    strategy (python/conflux.py pivoting enum; the C++ reference implements
    tournament only), restated as the parity target for the engine's
    no-pivot fast path (conflux_lu_set_pivoting(ctx, 0)).  Requires a
    diagonally dominant input.  Returns F: strict lower = L (unit diag
    implied), upper = U; the permutation is identity by construction."""
    import scipy.linalg as la
    N = A.shape[0]
    assert N % v == 0
    F = np.array(A, dtype=np.float64)
    for k0 in range(0, N, v):
        k1 = k0 + v
        T = F[k0:k1, k0:k1]
        for c in range(v):
            piv = T[c, c]
            if piv != 0.0:
                T[c + 1:, c] /= piv
            T[c + 1:, c + 1:] -= np.outer(T[c + 1:, c], T[c, c + 1:])
        if k1 < N:
            L = np.tril(T, -1) + np.eye(v)
            U = np.triu(T)
            F[k0:k1, k1:] = la.solve_triangular(
                L, F[k0:k1, k1:], lower=True, unit_diagonal=True)
            F[k1:, k0:k1] = la.solve_triangular(
                U, F[k1:, k0:k1].T, trans='T').T
            F[k1:, k1:] -= F[k1:, k0:k1] @ F[k0:k1, k1:]
    return F
