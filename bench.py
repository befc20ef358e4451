#!/usr/bin/env python3
"""bench.py — headline benchmark of the MI355X-native CONFLUX LU engine.

Measures BASELINE.json's metric — fp64 LU TFLOP/s, (2/3)N^3 / t_wall — on the
BASELINE configs.  A "step" is one full LU factorization of the synthetic
N x N fp64 matrix (the reference miniapp's repetition unit,
examples/conflux_miniapp.cpp:138-167).

    python bench.py [--gpus N] [--steps K] [--warmup W]

Single process when N==1; for N>1 the driver launches this file under
torch.distributed.run with one rank per GPU — torch.distributed (gloo) is
used ONLY to broadcast the RCCL unique id and for host barriers; all compute
and all data-path communication happen inside libconflux_lu.so (HIP + RCCL
over xGMI).

Workloads (grid restrictions Px==Py pow2):
    1 GPU : N=16384, v=512, grid 1x1x1   (BASELINE config 2)
    2 GPU : N=16384, v=512, grid 1x1x2   (depth replication)
    4 GPU : N=32768, v=512, grid 2x2x1   (BASELINE config 3)
    8 GPU : N=65536, v=512, grid 2x2x2   (BASELINE config 4)

Rank 0 prints ONE JSON line per the driver contract.
"""
import argparse
import json
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

GRIDS = {
    1: (16384, 512, 1, 1, 1),
    2: (16384, 512, 1, 1, 2),
    4: (32768, 512, 2, 2, 1),
    8: (65536, 512, 2, 2, 2),
}

FP64_MFMA_PEAK_TFLOPS = 78.6  # gfx950: 256 CU x 2.4 GHz x 128 f64 flop/clk/CU


def cpu_baseline():
    """Time the compiled reference (oracle/_ref, MKL+MPICH — kind
    'reference') on this box's host cores AT THE BENCH CONFIG's N (16384,
    v=512 — same size the GPU number beside it is quoted on; the matrix is
    2 GiB and fits host RAM).  Grid 4x4x1 under mpiexec -n 16 (the
    reference cannot run 1-wide grids; best measured rank/thread layout on
    this host class, tools/cpu_sweep.sh), one repetition (~20-30 s of CPU
    work).  Reported baseline, not target."""
    ref = os.path.join(os.path.dirname(os.path.abspath(__file__)), "oracle",
                       "_ref", "conflux_ref")
    if not os.path.exists(ref):
        return None
    ncores = os.cpu_count() or 1
    # best layout from the measured sweep on this host class
    # (tools/cpu_sweep.sh: 16 ranks x 8 OMP threads on a 4x4x1 grid beat
    #  4x64, 64x4, 16x16 and 8x32 at N=8192)
    n_ranks, omp, grid = 16, 8, ("4", "4", "1")
    if ncores < n_ranks * omp:
        n_ranks, omp = 4, max(1, ncores // 4)
        grid = ("2", "2", "1")
    env = dict(os.environ, MKL_THREADING_LAYER="GNU",
               OMP_NUM_THREADS=str(omp), LD_LIBRARY_PATH="/opt/conda/lib")
    N, v = 16384, 512
    try:
        out = subprocess.run(
            ["/opt/conda/bin/mpiexec", "-n", str(n_ranks), ref, str(N),
             str(v), grid[0], grid[1], grid[2], "-", "/tmp/confluxref_bench",
             "2"],
            env=env, capture_output=True, text=True, timeout=900)
        ms = None
        for line in out.stdout.splitlines():
            if line.startswith("_result_"):
                ms = float(line.split(",")[-2])
        if ms is None:
            return None
        tflops = (2.0 / 3.0) * N ** 3 / (ms * 1e-3) / 1e12
        return {"value": round(tflops, 4), "unit": "TFLOP/s",
                "cores": n_ranks * omp, "kind": "reference",
                "sample": f"reference CPU path (MKL+MPICH) N={N} v={v} "
                          f"grid {'x'.join(grid)}, {n_ranks} ranks x {omp} "
                          f"OMP threads, 1 rep on this box's host cores"}
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=None)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--N", type=int, default=None, help="override matrix dim")
    ap.add_argument("--tile", "--v", dest="v", type=int, default=None,
                    help="override tile size")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = args.gpus or world

    # pin this process to its GPU BEFORE the HIP runtime loads
    # (CONFLUX_BENCH_SHARE_GPU: all ranks on device 0 — only for the
    # shimccl rehearsal of this launch path on a 1-GPU box)
    if world > 1 and not os.environ.get("CONFLUX_BENCH_SHARE_GPU"):
        os.environ["HIP_VISIBLE_DEVICES"] = str(local_rank)

    N, v, Px, Py, Pz = GRIDS[n_gpus]
    K = None
    if args.N:
        N = args.N
    if args.v:
        v = args.v

    dist = None
    if world > 1:
        import torch.distributed as tdist
        tdist.init_process_group("gloo", rank=rank, world_size=world)
        dist = tdist

    from conflux_amd import Engine

    uid = None
    if world > 1:
        objs = [Engine.make_uid()] if rank == 0 else [None]
        dist.broadcast_object_list(objs, src=0)
        uid = objs[0]

    if os.environ.get("CONFLUX_BENCH_DRYRUN"):
        # bootstrap-only validation (CI on GPU-less hosts): arg parsing,
        # gloo init, uid broadcast — everything up to engine creation
        if rank == 0:
            print(json.dumps({"dryrun": True, "n_gpus": n_gpus, "N": N,
                              "v": v, "grid": f"{Px}x{Py}x{Pz}",
                              "uid_ok": uid is None or len(uid) == 128}))
        if dist:
            dist.barrier()
            dist.destroy_process_group()
        return

    eng = Engine(N, v, Px, Py, Pz, rank=(0 if world == 1 else rank),
                 world=world, uid=uid)
    eng.store_factors(False)  # timing mode (factor collection off, like the
    #                           reference's non-VALIDATION build)

    def barrier():
        if dist:
            dist.barrier()

    flops_per_step = (2.0 / 3.0) * N ** 3

    for _ in range(args.warmup):
        eng.init_matrix(42)
        eng.factor()
    barrier()
    t0 = time.perf_counter()
    engine_ms = []
    for _ in range(args.steps):
        eng.init_matrix(42)
        engine_ms.append(eng.factor())  # factor() ends with a device sync
    barrier()
    t1 = time.perf_counter()

    total_s = t1 - t0
    if dist:
        import torch
        tt = torch.tensor([total_s])
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        total_s = float(tt[0])

    stats = eng.kernel_stats()
    K = v // Pz  # nlayr: the trailing GEMM's K
    g = stats["dgemm_trailing"]
    roofline = None
    if g["launches"] > 0 and g["seconds"] > 0:
        # NOTE: with the 1-GPU lookahead overlap the trailing-GEMM launches
        # in the timed region run on a deliberately CAPPED grid (400/512
        # workgroup slots so the panel kernel keeps whole CUs) — `achieved`
        # therefore understates the kernel: it sustains 54.2 TF/s standalone
        # at this shape and 57.5-57.7 TF/s (73% of spec peak) at the
        # N=49152/65536 north-star shapes (tools/gemm_bench + DESIGN §5).
        achieved = g["flops"] / g["seconds"] / 1e12
        # Per-launch HBM bytes at the bench GEMM shape, PMC-measured
        # (rocprofv3 --pmc FETCH_SIZE / WRITE_SIZE in separate passes,
        # profiles/r02_pmc_dgemm.csv) and CALIBRATED on a known byte count
        # as the guide prescribes: WRITE_SIZE*1024 / exact C-write bytes
        # (N^2*8) = 1.4531 at both measured shapes, so raw (FETCH+WRITE)
        # KB * 1024 / 1.4531.  Result: ~1.0x the algorithmic A+B+2C bytes
        # at 16384 and ~1.13x at 49152 (strip-ordered tiles; r01's "3x"
        # was the misapplied x2 wide-read correction + flat tile order).
        # keyed by (global N, nlayr); the value is the measured per-launch
        # traffic of THAT config's dominant per-rank GEMM shape
        traffic = {
            (16384, 512): 4.33e9,   # 1 GPU: launch 16384^2 K=512 (0.98x alg)
            (49152, 512): 4.43e10,  # 1-GPU cfg-4-scale: 49152^2 K=512 (1.13x)
            (32768, 512): 4.33e9,   # cfg 3 (4 GPU): per-rank 16384^2 K=512
            (65536, 256): 1.59e10,  # cfg 4 (8 GPU): per-rank 32768^2 K=256
                                    # (0.91x algorithmic)
        }.get((N, K))
        roofline = {
            "bound": "mfma",
            "achieved": round(achieved, 3),
            "peak": FP64_MFMA_PEAK_TFLOPS,
            "unit": "TFLOP/s",
            "frac": round(achieved / FP64_MFMA_PEAK_TFLOPS, 4),
            "traffic": traffic,
        }

    eng.close()

    if rank == 0:
        tflops = args.steps * flops_per_step / total_s / 1e12
        result = {
            "metric": "fp64_lu_tflops",
            "value": round(tflops, 3),
            "unit": "TFLOP/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(total_s * 1e3 / args.steps, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": f"conflux LU N={N} v={v} grid {Px}x{Py}x{Pz} "
                            f"(BASELINE cfg for {n_gpus} GPU)",
                "N": N, "v": v, "grid": f"{Px}x{Py}x{Pz}",
            },
            "engine_ms_per_step": [round(x, 1) for x in engine_ms],
            "kernel_stats": {k: {"seconds": round(s["seconds"], 4),
                                 "launches": s["launches"]}
                             for k, s in stats.items()},
        }
        if roofline:
            result["roofline"] = roofline
        if n_gpus == 1 and not args.skip_cpu_baseline:
            cb = cpu_baseline()
            if cb:
                result["cpu_baseline"] = cb
        print(json.dumps(result))

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
