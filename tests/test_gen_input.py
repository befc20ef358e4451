import numpy as np

from oracle.gen_input import gen_block, gen_matrix


def test_deterministic():
    a = gen_matrix(32)
    b = gen_matrix(32)
    assert np.array_equal(a, b)


def test_block_consistency():
    """Any sub-block equals the same slice of the full matrix (grid-independent)."""
    A = gen_matrix(64)
    blk = gen_block(16, 48, 8, 40, 64)
    assert np.array_equal(A[16:48, 8:40], blk)


def test_range_and_spread():
    A = gen_matrix(128)
    assert A.min() >= 5.0 and A.max() < 6.0
    assert abs(A.mean() - 5.5) < 0.01


def test_seed_changes_matrix():
    assert not np.array_equal(gen_matrix(16, seed=42), gen_matrix(16, seed=43))
