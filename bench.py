#!/usr/bin/env python3
"""Flagship benchmark: distributed LM+PCG bundle adjustment on a synthetic
BAL Venice-1778-shaped problem (BASELINE.json headline config: 1778 cameras,
993923 points, ~5M observations, fp64).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`, launched
via torch.distributed.run for N>1 (one rank per GPU over RCCL).  One "step" =
one full LM iteration (forward + Jacobians, Hessian assembly + allreduce,
damped Schur-complement PCG with its per-iteration allreduces, parameter
update, gain-ratio accept/reject) — the unit the reference logs per
iteration (MegBA lm_algo.cu "Iter k ... elapsed ms").  Data is synthetic
(no network for datasets); parameters are ground truth + noise so LM runs a
realistic trajectory; force_iterations keeps the step count exact.
"""
import argparse
import json
import os
import re
import subprocess
import sys
import threading
import time

MODELS = {
    # BASELINE.json configs (synthetic, same shapes as the BAL datasets).
    "venice1778": dict(ncam=1778, npt=993923, nobs=5_000_000),
    "trafalgar257": dict(ncam=257, npt=65132, nobs=225_911),
    "ladybug49": dict(ncam=49, npt=7776, nobs=31_843),
    "final13682": dict(ncam=13682, npt=4_456_117, nobs=28_987_644),
    "synth20k": dict(ncam=20_000, npt=10_000_000, nobs=50_000_000),
    "tiny": dict(ncam=30, npt=400, nobs=3_000),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="venice1778", choices=sorted(MODELS))
    ap.add_argument("--device", default="gpu", choices=["gpu", "cpu"])
    ap.add_argument("--dtype", default="float64",
                    choices=["float64", "float32"])
    ap.add_argument("--diff", default="auto", choices=["auto", "analytical"])
    # implicit (matrix-free Schur products over packed vector-group J)
    # measures FASTER than explicit on MI355X (57 vs ~80 ms/step on
    # Venice fp64 after the r2 packed/banded rework, profiles/) and uses
    # ~30% less memory; same LM+PCG math, same fixed work per step.  The
    # reference ships both (BAL_Double / BAL_Double_implicit).
    ap.add_argument("--schur", default="implicit",
                    choices=["explicit", "implicit"])
    ap.add_argument("--verbose", action="store_true")
    args = ap.parse_args()
    if args.steps < 1:
        ap.error("--steps must be >= 1")
    if args.warmup < 0:
        ap.error("--warmup must be >= 0")

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    import megba_amd as mb

    shape = MODELS[args.model]
    t0 = time.time()
    cams, pts, ci, pi, meas = mb.synthesize_bal(
        shape["ncam"], shape["npt"], shape["nobs"], seed=7)
    if rank == 0:
        print(f"# synthesized {args.model} in {time.time()-t0:.1f}s",
              file=sys.stderr)

    dist = None
    store = None
    rccl_id = None
    allreduce = None
    launched_distributed = ("TORCHELASTIC_RUN_ID" in os.environ or
                            ("MASTER_ADDR" in os.environ and
                             "MASTER_PORT" in os.environ and
                             "RANK" in os.environ))
    if world > 1 or launched_distributed:
        if args.device == "gpu":
            # Torch-free bootstrap: a stdlib-TCP store exchanges the
            # 128-byte ncclUniqueId and serves barriers / the max-over-
            # ranks reduction; RCCL then runs natively over xGMI.
            from megba_amd import _core
            from megba_amd.rendezvous import from_env
            store = from_env(rank, world)
            ngpu = max(1, _core.hip_device_count())
            dev = local_rank % ngpu
            if world > 1 and not os.environ.get("MEGBA_NO_PREFLIGHT"):
                # Pre-flight: throwaway comm + 1-element allreduce under a
                # watchdog, so a wedged RCCL bootstrap aborts the whole job
                # with a clear message instead of hanging the scale run.
                pf_id = store.broadcast_bytes(
                    _core.rccl_unique_id() if rank == 0 else None)
                try:
                    t_pf = _core.rccl_preflight(pf_id, rank, world, dev,
                                                120.0)
                except Exception as e:
                    print(f"RCCL BOOTSTRAP FAILED (rank {rank}/{world}, "
                          f"device {dev}): {e}", file=sys.stderr, flush=True)
                    sys.exit(1)
                if rank == 0:
                    print(f"# rccl preflight ok ({t_pf:.2f}s, world={world})",
                          file=sys.stderr)
            rccl_id = store.broadcast_bytes(
                _core.rccl_unique_id() if rank == 0 else None)
        else:
            # multi-process CPU testing path: gloo allreduce callback
            import torch.distributed as tdist
            dist = tdist
            dist.init_process_group("gloo", rank=rank, world_size=world)
            from megba_amd.dist import gloo_allreduce_callback
            allreduce = gloo_allreduce_callback()

    device_index = local_rank
    if args.device == "gpu":
        from megba_amd import _core
        ngpu = max(1, _core.hip_device_count())
        device_index = local_rank % ngpu
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device=args.device, dtype=args.dtype, rank=rank,
            world_size=world, device_index=device_index, diff=args.diff,
            schur=args.schur, allreduce=allreduce, rccl_id=rccl_id)

    # Fixed-work steps: every LM iteration runs the full solver_max_iter=100
    # PCG iterations (the reference demo's budget, README.md:54-67) with the
    # tol/refuse early exits disabled.  The real solver honours tol/refuse
    # (see examples/bal_solve.py); here they are disabled so each timed step
    # does identical work -- trajectory-dependent early exits otherwise make
    # ms/step incomparable across runs/ranks/world sizes.  This is strictly
    # MORE work per step than the reference demo does.
    p.lm_init(tau=1e4, epsilon1=1.0, epsilon2=1e-10, solver_max_iter=100,
              solver_tol=0.0, solver_refuse_ratio=1e30,
              force_iterations=True, verbose=args.verbose and rank == 0)

    def sync():
        if args.device == "gpu":
            from megba_amd import _core
            if _core.hip_device_count() > 0:
                _core.device_synchronize()
        if store is not None:
            store.barrier()
        if dist is not None:
            dist.barrier()

    # Self-sampled GPU-utilization corroboration (a single
    # external SMI sample missed the 1-2 s timed region): a background
    # thread polls rocm-smi during the timed loop; mean/max go into the
    # JSON line as extra fields.
    busy_samples = []
    stop_sampling = threading.Event()

    def _sample():
        while not stop_sampling.is_set():
            try:
                out = subprocess.run(
                    ["rocm-smi", "--showuse"], capture_output=True,
                    text=True, timeout=5).stdout
                vals = [float(m) for m in
                        re.findall(r"GPU use \(%\)\s*:\s*([0-9.]+)", out)]
                if vals:
                    busy_samples.append(max(vals))
            except Exception:
                return
            stop_sampling.wait(0.2)

    for _ in range(args.warmup):
        p.lm_step()
    sync()
    sampler = None
    if rank == 0 and args.device == "gpu":
        sampler = threading.Thread(target=_sample, daemon=True)
        sampler.start()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        log = p.lm_step()
    sync()
    elapsed = time.perf_counter() - t0
    stop_sampling.set()
    if sampler is not None:
        sampler.join(timeout=2)

    # MAX over ranks.
    if store is not None:
        elapsed = store.all_max(elapsed)
    elif dist is not None:
        import torch
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000.0
        line = {
            "metric": "lm_iterations_per_s",
            "value": args.steps / elapsed,
            "unit": "iters/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64" if args.dtype == "float64" else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"BAL-{args.model}-synthetic",
                "cameras": shape["ncam"],
                "points": shape["npt"],
                "observations": shape["nobs"],
                "diff": args.diff,
                "schur": args.schur,
                "parallelism": f"edge-dp{world}",
                "solver": {"tau": 1e4, "solver_tol": 0.0,
                           "solver_refuse_ratio": 1e30,
                           "solver_max_iter": 100,
                           "note": "fixed-work steps: 100 PCG iters/step"},
                "final_chi2": log["chi2"],
            },
        }
        if busy_samples:
            line["gpu_busy_self"] = {
                "samples": len(busy_samples),
                "mean": round(sum(busy_samples) / len(busy_samples), 1),
                "max": round(max(busy_samples), 1),
            }
        print(json.dumps(line), flush=True)

    if store is not None:
        store.barrier()  # keep rank 0 alive until every rank has printed
        store.close()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
