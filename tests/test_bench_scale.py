"""Bench-scale correctness: the sizes the bench lines are quoted on
self-check every round (correctness evidence previously
stopped at N=4096 while the bench ran N>=16384 timing-only).

One store_factors factorization per bench size through the miniapp CLI
(factor collection on, like the reference's CONFLUX_WITH_VALIDATION build),
asserting the device-side stripe-streamed ||PA-LU||_F/||A||_F residual.
"""
import os
import re
import subprocess

import pytest

REPO = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..")
MINIAPP = os.path.join(REPO, "conflux_amd", "conflux_miniapp")

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("N,v,timeout", [
    (16384, 512, 900),    # BASELINE cfg 2 (the 1-GPU bench line)
    (65536, 512, 1800),   # BASELINE cfg 4 size on one GPU (README large-N
                          # datapoint; exercises the >2^31-element launches)
])
def test_single_gpu_validated(N, v, timeout):
    if not os.path.exists(MINIAPP):
        pytest.skip("conflux_miniapp not built")
    out = subprocess.run(
        [MINIAPP, "-N", str(N), "-b", str(v), "--p_grid=1,1,1", "-r", "1"],
        capture_output=True, text=True, timeout=timeout)
    assert out.returncode == 0, out.stdout + out.stderr
    m = re.search(r"relative residual \|\|PA-LU\|\|_F/\|\|A\|\|_F = (\S+)",
                  out.stdout)
    assert m, out.stdout
    assert float(m.group(1)) < 1e-13


def test_cholesky_single_rank_validated():
    """The non-sim single-rank Cholesky path (lookahead chain on the second
    stream + adaptive masked c4 rectangle) validates on device."""
    chol = os.path.join(REPO, "conflux_amd", "cholesky_miniapp")
    if not os.path.exists(chol):
        pytest.skip("cholesky_miniapp not built")
    out = subprocess.run(
        [chol, "--dim", "4096", "--tile", "512", "--grid", "1,1,1",
         "--run", "1"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stdout + out.stderr
    m = re.search(r"relative residual \|\|A-LL\^T\|\|_F/\|\|A\|\|_F = (\S+)",
                  out.stdout)
    assert m, out.stdout
    assert float(m.group(1)) < 1e-13


def test_miniapp_nopivot_validated():
    """CLI surface of the no-pivot fast path: --pivoting none factors the
    diagonally dominant SPD fill and the device residual validates it."""
    if not os.path.exists(MINIAPP):
        pytest.skip("conflux_miniapp not built")
    out = subprocess.run(
        [MINIAPP, "-N", "2048", "-b", "256", "--p_grid=1,1,1", "-r", "1",
         "--pivoting", "none"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    m = re.search(r"relative residual \|\|PA-LU\|\|_F/\|\|A\|\|_F = (\S+)",
                  out.stdout)
    assert m, out.stdout
    assert float(m.group(1)) < 1e-14
