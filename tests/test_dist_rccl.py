"""Distributed-path tests: the engine's REAL !sim branches executed as
multi-process runs on ONE GPU through the shimccl mock transport
(tests/shimccl.cpp).  Real RCCL refuses two ranks on one device, so this is
the only way to EXECUTE the grouped send/recv choreography, the dual-comm
lookahead and the distributed validation before a multi-GPU node exists.
The transport mock preserves the NCCL semantics the engine relies on
(per-pair FIFO matching, group semantics, size-mismatch = loud error), so a
green run here means the choreography — counts, offsets, pairing, ordering —
is right; only RCCL itself remains untested until a multi-GPU run.

Parity bar (same as the sim-mode tests): pivots bit-exact vs the oracle,
factors <= 1e-11, distributed residual <= 1e-13, identical across ranks.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.join(HERE, "..")
SHIM = os.path.join(HERE, "shimccl.so")
MINIAPP = os.path.join(REPO, "conflux_amd", "conflux_miniapp")

pytestmark = pytest.mark.gpu


def _dist_env(tmp_path, async_mode=False):
    env = dict(os.environ)
    env["LD_PRELOAD"] = SHIM
    env["SHIMCCL_DIR"] = str(tmp_path / "box")
    env.pop("HIP_VISIBLE_DEVICES", None)  # all ranks share device 0
    if async_mode:
        # stream-enqueued transport (hipLaunchHostFunc): real-RCCL
        # completion semantics — the host never blocks in a comm call
        env["SHIMCCL_ASYNC"] = "1"
    return env


def _run_ranks(tmp_path, N, v, Px, Py, Pz, reps=1, timeout=600,
               set_matrix=False, async_mode=False, mode=None):
    if not os.path.exists(SHIM):
        pytest.skip("shimccl.so not built (make -C tests)")
    P = Px * Py * Pz
    env = _dist_env(tmp_path, async_mode)
    procs, outs = [], []
    for r in range(P):
        out = str(tmp_path / f"rank{r}.npz")
        outs.append(out)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(HERE, "dist_worker.py"),
             str(N), str(v), str(Px), str(Py), str(Pz), str(r), str(reps),
             out] + (["set_matrix"] if set_matrix else [mode] if mode
                     else []),
            env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            text=True))
    logs = []
    try:
        for p in procs:
            o, _ = p.communicate(timeout=timeout)
            logs.append(o)
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
    for r, (p, log) in enumerate(zip(procs, logs)):
        assert p.returncode == 0, f"rank {r} failed:\n{log}"
    return [np.load(o) for o in outs]


def _assemble_F(results, N, v, Px, Py, Pz):
    """Reassemble global F from the pk == 0 ranks' tile-cyclic locals
    (owner map layout.cpp:95-123)."""
    F = np.zeros((N, N))
    for pi in range(Px):
        for pj in range(Py):
            g = (pi * Py + pj) * Pz + 0
            loc = results[g]["F"]
            for lti in range(N // (v * Px)):
                for ltj in range(N // (v * Py)):
                    gti, gtj = lti * Px + pi, ltj * Py + pj
                    F[gti * v:(gti + 1) * v, gtj * v:(gtj + 1) * v] = \
                        loc[lti * v:(lti + 1) * v, ltj * v:(ltj + 1) * v]
    return F


@pytest.mark.parametrize("grid,N,v,reps,ftol", [
    ((1, 1, 2), 1024, 128, 1, 1e-11),  # depth reduce + gpivots broadcast
    ((2, 2, 1), 1024, 128, 2, 1e-11),  # butterfly, A00 exchange, spreads
    ((2, 2, 2), 1024, 128, 1, 1e-11),  # the full 3D choreography
    ((2, 2, 1), 4096, 512, 1, 1e-9),   # BASELINE tile size: v=512 panels,
                                       # tournament across 2 rank rows
                                       # (larger N -> looser element tol;
                                       # pivots stay bit-exact)
    ((4, 4, 1), 2048, 128, 1, 1e-10),  # 16 ranks: 2-round butterfly and
                                       # 4-row pivot routing on the real
                                       # dist code
])
def test_dist_parity_vs_oracle(tmp_path, grid, N, v, reps, ftol):
    from oracle import Params, gen_matrix, lu_oracle

    Px, Py, Pz = grid
    results = _run_ranks(tmp_path, N, v, Px, Py, Pz, reps=reps)

    A = gen_matrix(N)
    ref = lu_oracle(A, Params(N, v, Px, Py, Pz))

    # every rank reports the same pivots, bit-exact vs the oracle
    for r, res in enumerate(results):
        assert np.array_equal(res["perm"], ref["perm"]), f"rank {r} pivots"
    # every rank returns the identical broadcast residual, and it is small
    resids = [float(res["resid"]) for res in results]
    assert max(resids) == min(resids)
    assert max(resids) < 1e-13
    # factors match the oracle
    F = _assemble_F(results, N, v, Px, Py, Pz)
    assert np.abs(F - ref["F"]).max() < ftol
    # ... and are BIT-IDENTICAL to simulation mode: the sim and distributed
    # paths share every kernel and the deterministic pk-ascending combine
    # order; only the transport differs, so any divergence is a transport
    # ordering bug
    from conflux_amd import Engine
    with Engine(N, v, Px, Py, Pz, rank=-1) as e:
        e.store_factors(True)
        e.init_matrix(42)
        e.factor()
        Fsim = e.get_F_global()
        perm_sim = e.get_perm()
    assert np.array_equal(perm_sim, results[0]["perm"])
    assert np.array_equal(Fsim, F), "sim and distributed factors must be bit-identical"


@pytest.mark.parametrize("grid,N,v", [
    ((1, 1, 2), 1024, 128),
    ((2, 2, 2), 1024, 128),
])
def test_dist_parity_async_transport(tmp_path, grid, N, v):
    """Same choreography under SHIMCCL_ASYNC=1 — stream-enqueued transport
    with real-RCCL completion semantics (the host never blocks inside a
    comm call), so the dual-comm lookahead and event gating run with true
    asynchronous ordering.  Results must stay bit-exact."""
    from oracle import Params, gen_matrix, lu_oracle

    Px, Py, Pz = grid
    results = _run_ranks(tmp_path, N, v, Px, Py, Pz, reps=2,
                         async_mode=True)
    ref = lu_oracle(gen_matrix(N), Params(N, v, Px, Py, Pz))
    for r, res in enumerate(results):
        assert np.array_equal(res["perm"], ref["perm"]), f"rank {r} pivots"
    resids = [float(res["resid"]) for res in results]
    assert max(resids) == min(resids) and max(resids) < 1e-13
    F = _assemble_F(results, N, v, Px, Py, Pz)
    assert np.abs(F - ref["F"]).max() < 1e-11


def test_dist_set_matrix_local(tmp_path):
    """The documented drop-in data path (caller-supplied tile-cyclic local
    slices through conflux_lu_set_matrix_local) in a real multi-process
    world: same matrix as the generator, so the same oracle pins it."""
    from oracle import Params, gen_matrix, lu_oracle

    N, v, Px, Py, Pz = 1024, 128, 2, 2, 1
    results = _run_ranks(tmp_path, N, v, Px, Py, Pz, set_matrix=True)
    ref = lu_oracle(gen_matrix(N), Params(N, v, Px, Py, Pz))
    for res in results:
        assert np.array_equal(res["perm"], ref["perm"])
    F = _assemble_F(results, N, v, Px, Py, Pz)
    assert np.abs(F - ref["F"]).max() < 1e-11


def test_dist_nopivot(tmp_path):
    """The no-pivot fast path through the real multi-process
    distributed branches: identity permutation, factors vs the numpy
    no-pivot restatement on the SPD (diagonally dominant) generator fill."""
    from oracle import gen_matrix, lu_nopivot

    N, v, Px, Py, Pz = 1024, 128, 2, 2, 2
    results = _run_ranks(tmp_path, N, v, Px, Py, Pz, mode="nopiv")
    A = gen_matrix(N)
    S = 0.5 * (A + A.T) + 2.0 * N * np.eye(N)
    ref = lu_nopivot(S, v)
    for res in results:
        assert np.array_equal(res["perm"], np.arange(N))
        assert float(res["resid"]) < 1e-14
    F = _assemble_F(results, N, v, Px, Py, Pz)
    assert np.abs(F - ref).max() < 1e-11 * np.abs(ref).max()


def test_bench_dist_launch(tmp_path):
    """Rehearse the EXACT multi-rank bench launch the driver uses at round
    end (torch.distributed.run, one rank per GPU) on one GPU via shimccl:
    gloo bootstrap, uid broadcast, distributed engine, barriers, the MAX
    reduction, and the JSON contract line."""
    import json
    if not os.path.exists(SHIM):
        pytest.skip("shimccl.so not built (make -C tests)")
    env = _dist_env(tmp_path)
    env["CONFLUX_BENCH_SHARE_GPU"] = "1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29532", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--N", "1024",
         "--tile", "128", "--skip-cpu-baseline"],
        env=env, capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stdout + out.stderr
    lines = [l for l in out.stdout.splitlines() if l.startswith('{"metric"')]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["config"]["grid"] == "1x1x2"
    assert d["value"] > 0 and d["ms_per_step"] > 0


@pytest.mark.parametrize("grid,async_mode", [
    ("2,2,1", False), ("2,2,2", False), ("2,2,2", True),
])
def test_cholesky_dist(tmp_path, grid, async_mode):
    """The CONFCHOX distributed branches (depth reduce, L_kk column
    broadcast, slab + transpose spreads) executed multi-process through
    shimccl — also under the async stream-enqueued transport — with the
    distributed Cholesky validation."""
    chol = os.path.join(REPO, "conflux_amd", "cholesky_miniapp")
    if not os.path.exists(SHIM):
        pytest.skip("shimccl.so not built (make -C tests)")
    if not os.path.exists(chol):
        pytest.skip("cholesky_miniapp not built")
    env = _dist_env(tmp_path, async_mode)
    env["CONFLUX_SPAWN_OVERSUBSCRIBE"] = "1"
    out = subprocess.run(
        [chol, "--dim", "2048", "--tile", "256", "--grid", grid,
         "--run", "1"],
        env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    resid = [l for l in out.stdout.splitlines() if "relative residual" in l]
    assert resid, out.stdout
    assert float(resid[0].split("=")[-1]) < 1e-13


def test_miniapp_selfspawn_dist(tmp_path):
    """The CLI self-spawn launcher end to end: fork+exec per rank, uid file
    handoff, distributed factor + validate, `_result_` contract."""
    if not os.path.exists(SHIM):
        pytest.skip("shimccl.so not built (make -C tests)")
    if not os.path.exists(MINIAPP):
        pytest.skip("conflux_miniapp not built")
    env = _dist_env(tmp_path)
    env["CONFLUX_SPAWN_OVERSUBSCRIBE"] = "1"
    out = subprocess.run(
        [MINIAPP, "-N", "1024", "-b", "128", "--p_grid=2,2,1", "-r", "1"],
        env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    res = [l for l in out.stdout.splitlines()
           if l.startswith("_result_ lu,conflux,1")]  # data line, not header
    assert len(res) == 1, out.stdout
    parts = res[0].split(",")
    assert parts[2] == "1024" and parts[4] == "4" and parts[5] == "2x2x1"
    resid = [l for l in out.stdout.splitlines()
             if "relative residual" in l]
    assert resid, out.stdout
    assert float(resid[0].split("=")[-1]) < 1e-13
