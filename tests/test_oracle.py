"""Pin the numpy oracle.

1) On the 1x1x1 grid the reference algorithm degenerates to standard blocked
   right-looking LU with partial pivoting, so the oracle must reproduce
   LAPACK (scipy.linalg.lu_factor) pivots BIT-EXACTLY and its packed factors
   to rounding.  (The reference binary itself crashes/NaNs on Px=1 — latent
   A00Buff gap, conflux_opt.hpp:778+308 — so LAPACK is the 1-wide pin.)
2) On Px >= 2 grids the oracle must match the compiled reference
   (oracle/_ref, golden fixtures from tests/golden/make_golden.py):
   pivot indices bit-exact, factored matrix to fp64 rounding.
"""
import numpy as np
import pytest
import scipy.linalg as la

from oracle import Params, gen_matrix, lu_oracle, residual_check

TOL_FACTORS = 1e-11       # |F_oracle - F_ref|_max on [5,6)-valued inputs
TOL_RESID = 1e-14         # ||PA-LU||/||A|| at these sizes


@pytest.mark.parametrize("N,v", [(32, 4), (64, 8), (64, 16), (128, 32), (256, 64)])
def test_1x1x1_matches_lapack(N, v):
    A = gen_matrix(N)
    r = lu_oracle(A, Params(N, v, 1, 1, 1))
    lu, piv = la.lu_factor(A)
    perm = np.arange(N)
    for i, j in enumerate(piv):
        perm[i], perm[j] = perm[j], perm[i]
    assert np.array_equal(r["perm"], perm), "pivot indices must be bit-exact"
    assert np.abs(r["F"] - lu).max() < TOL_FACTORS
    assert residual_check(A, r["perm"], r["F"]) < TOL_RESID


def _tags(g):
    return sorted({k.split("/")[0] for k in g.files})


def test_golden_parity(golden):
    checked = 0
    for tag in _tags(golden):
        N, v, Px, Py, Pz = (int(x) for x in golden[f"{tag}/cfg"])
        A = golden[f"{tag}/A"]
        C_ref = golden[f"{tag}/C"]
        perm_ref = golden[f"{tag}/perm"]
        r = lu_oracle(A, Params(N, v, Px, Py, Pz))
        assert np.array_equal(r["perm"], perm_ref), f"{tag}: pivots differ"
        assert np.abs(r["F"] - C_ref).max() < TOL_FACTORS, f"{tag}: factors differ"
        assert residual_check(A, perm_ref, C_ref) < TOL_RESID
        checked += 1
    assert checked >= 10


def test_kat_planted_pivot(golden):
    """The N=16 KAT plants a 900 at global row 5, column 2
    (lu_params.hpp:204) owned by a different rank row than the diagonal —
    the tournament must pick it when column 2 is eliminated."""
    perm = golden["kat16_v4_221/perm"]
    assert 5 in perm[:8]  # row 5 chosen within the first two panels


@pytest.mark.parametrize("grid", [(2, 2, 1), (2, 2, 2), (4, 4, 2)])
def test_residual_random(grid):
    Px, Py, Pz = grid
    N, v = 128, 8
    A = gen_matrix(N)
    r = lu_oracle(A, Params(N, v, Px, Py, Pz))
    assert residual_check(A, r["perm"], r["F"]) < TOL_RESID
    # every row used exactly once
    assert np.array_equal(np.sort(r["perm"]), np.arange(N))


@pytest.mark.filterwarnings("ignore::scipy.linalg.LinAlgWarning")
def test_zero_pivot_column_matches_lapack():
    """A singular input follows LAPACK's dgetrf convention: the zero pivot
    skips the scaling (dscal is not applied when the pivot is 0) and the
    factorization continues.  The zero column sits in the LAST tile column:
    a zero pivot in an earlier panel makes BLOCKED LU ill-defined
    downstream (the A10*U^-1 TRSM divides by the zero diagonal — true for
    the reference's cblas_dtrsm and LAPACK's blocked dgetrf alike)."""
    import numpy as np
    import scipy.linalg as la
    N, v = 64, 32
    A = gen_matrix(N)
    A[:, 40] = 0.0
    r = lu_oracle(A, Params(N, v, 1, 1, 1))
    lu, piv = la.lu_factor(A)
    ref = np.arange(N)
    for i, j in enumerate(piv):
        ref[i], ref[j] = ref[j], ref[i]
    assert np.array_equal(r["perm"], ref)
