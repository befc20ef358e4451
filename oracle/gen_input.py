"""Deterministic, grid-independent synthetic input generator.

The reference fills the matrix per rank with mt19937_64(seed + rank)
sequential draws (lu_params.hpp:364-375), which makes the *global* matrix
depend on the process grid and on COSTA's traversal order. As sanctioned by
SURVEY.md §8(d), we deviate: A[i, j] = 5 + U[0,1) where the uniform draw is a
counter-based splitmix64 hash of (seed, i, j). The same function is
implemented bit-identically in the C++ engine (conflux_amd/csrc/engine.cpp)
and in the compiled-reference driver, so every path factors the same matrix
for any grid.

Entries lie in [5, 6) like the reference's fill, keeping dgetrf
well-conditioned at the bench sizes.
"""
import numpy as np

_GAMMA = np.uint64(0x9E3779B97F4A7C15)
_M1 = np.uint64(0xBF58476D1CE4E5B9)
_M2 = np.uint64(0x94D049BB133111EB)


def _splitmix64(z: np.ndarray) -> np.ndarray:
    z = (z + _GAMMA).astype(np.uint64)
    z ^= z >> np.uint64(30)
    z *= _M1
    z ^= z >> np.uint64(27)
    z *= _M2
    z ^= z >> np.uint64(31)
    return z


def gen_block(i0: int, i1: int, j0: int, j1: int, N: int, seed: int = 42) -> np.ndarray:
    """A[i0:i1, j0:j1] of the global N x N matrix, fp64."""
    with np.errstate(over="ignore"):
        ii = np.arange(i0, i1, dtype=np.uint64)[:, None]
        jj = np.arange(j0, j1, dtype=np.uint64)[None, :]
        key = (ii << np.uint64(32)) ^ jj
        key = key + np.uint64(seed) * _M1
        h = _splitmix64(key)
    u = (h >> np.uint64(11)).astype(np.float64) * (1.0 / 9007199254740992.0)
    return 5.0 + u


def gen_matrix(N: int, seed: int = 42) -> np.ndarray:
    return gen_block(0, N, 0, N, N, seed)
