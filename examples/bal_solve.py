#!/usr/bin/env python3
"""BAL solver CLI -- the equivalent of the reference's BAL_Double /
BAL_Double_analytical / BAL_*_implicit binaries (same flags, same semantics;
/root/reference/examples/BAL_Double.cpp:50-58).

Examples:
  # reference demo invocation on a BAL text file (Venice-1778):
  python examples/bal_solve.py --path problem-1778-993923-pre.txt \
      --world_size 1 --max_iter 100 --solver_tol 1e-1 \
      --solver_refuse_ratio 1 --solver_max_iter 100 --tau 1e4 \
      --epsilon1 1 --epsilon2 1e-10

  # no dataset available: run on a synthetic Venice-shaped problem:
  python examples/bal_solve.py --synthetic venice \
      --world_size 1 --max_iter 20

Multi-GPU: launch with torch.distributed.run (one rank per GPU over RCCL):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/bal_solve.py --path ... --world_size 8
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

SYNTH = {
    "ladybug": (49, 7776, 31843),
    "trafalgar": (257, 65132, 225911),
    "venice": (1778, 993923, 5000000),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--path", default="", help="BAL problem-*.txt file")
    ap.add_argument("--synthetic", default="", choices=[""] + sorted(SYNTH),
                    help="use a synthetic problem of this BAL shape")
    ap.add_argument("--world_size", type=int, default=1)
    ap.add_argument("--max_iter", type=int, default=20)
    ap.add_argument("--solver_max_iter", type=int, default=50)
    ap.add_argument("--solver_tol", type=float, default=10.0)
    ap.add_argument("--solver_refuse_ratio", type=float, default=1.0)
    ap.add_argument("--tau", type=float, default=1.0)
    ap.add_argument("--epsilon1", type=float, default=1.0)
    ap.add_argument("--epsilon2", type=float, default=1e-10)
    ap.add_argument("--device", default="gpu", choices=["gpu", "cpu"])
    ap.add_argument("--dtype", default="float64", choices=["float64", "float32"])
    ap.add_argument("--diff", default="auto", choices=["auto", "analytical"])
    ap.add_argument("--schur", default="explicit", choices=["explicit", "implicit"])
    ap.add_argument("--loss", default="none", choices=["none", "huber", "cauchy"])
    ap.add_argument("--loss_delta", type=float, default=1.0)
    ap.add_argument("--out", default="", help="write solved BAL file here")
    args = ap.parse_args()

    import megba_amd as mb

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.world_size)))

    if args.path:
        cams, pts, ci, pi, meas = mb.load_bal(args.path)
    elif args.synthetic:
        cams, pts, ci, pi, meas = mb.synthesize_bal(*SYNTH[args.synthetic], seed=7)
    else:
        ap.error("need --path or --synthetic")

    print(f"solving {args.path or args.synthetic}, world_size: {world}, "
          f"max iter: {args.max_iter}, solver_tol: {args.solver_tol}, "
          f"solver_refuse_ratio: {args.solver_refuse_ratio}, "
          f"solver_max_iter: {args.solver_max_iter}, tau: {args.tau}, "
          f"epsilon1: {args.epsilon1}, epsilon2: {args.epsilon2}")

    allreduce = None
    rccl_id = None
    if world > 1:
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        if args.device == "gpu":
            from megba_amd.dist import broadcast_rccl_id
            rccl_id = broadcast_rccl_id(rank)
        else:
            from megba_amd.dist import gloo_allreduce_callback
            allreduce = gloo_allreduce_callback()

    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device=args.device, dtype=args.dtype, rank=rank, world_size=world,
            device_index=int(os.environ.get("LOCAL_RANK", rank)),
            diff=args.diff, schur=args.schur, loss=args.loss,
            loss_delta=args.loss_delta, allreduce=allreduce,
            rccl_id=rccl_id)
    p.solve(max_iter=args.max_iter, tau=args.tau, epsilon1=args.epsilon1,
            epsilon2=args.epsilon2, solver_max_iter=args.solver_max_iter,
            solver_tol=args.solver_tol,
            solver_refuse_ratio=args.solver_refuse_ratio,
            verbose=rank == 0)
    if args.out and rank == 0:
        c2, p2 = p.get_params()
        mb.save_bal(args.out, c2, p2, ci, pi, meas)
        print("wrote", args.out)


if __name__ == "__main__":
    main()
