"""bench.py multi-rank bootstrap, CPU-only (CONFLUX_BENCH_DRYRUN): the
exact launch the driver uses for the N>1 scaling bench — torch.distributed.run
with one rank per GPU — through arg parsing, gloo init, REAL-RCCL unique-id
generation on rank 0 and its broadcast, stopping just before engine
creation.  Keeps the never-run-before-round-end path honest on every CI
run."""
import json
import os
import subprocess
import sys

REPO = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..")


def test_bench_dryrun_world2():
    env = dict(os.environ, CONFLUX_BENCH_DRYRUN="1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", os.path.join(REPO, "bench.py"),
         "--gpus", "2"],
        env=env, capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stdout + out.stderr
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout
    d = json.loads(lines[0])
    assert d == {"dryrun": True, "n_gpus": 2, "N": 16384, "v": 512,
                 "grid": "1x1x2", "uid_ok": True}
