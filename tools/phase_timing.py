#!/usr/bin/env python3
"""Per-phase wall timing of one LM iteration on the GPU (run via gpurun).

Separates: forward, assembly (build_linear_system incl. transpose+allreduce),
processDiag, PCG solve (and per-PCG-iteration cost), norms, update, rho.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import megba_amd as mb


def t(fn, reps=5):
    fn()  # warm
    best = 1e9
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        best = min(best, time.perf_counter() - t0)
    return best * 1000


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="venice1778")
    ap.add_argument("--diff", default="auto")
    ap.add_argument("--schur", default="explicit")
    args = ap.parse_args()
    from bench import MODELS
    shape = MODELS[args.model]
    cams, pts, ci, pi, meas = mb.synthesize_bal(
        shape["ncam"], shape["npt"], shape["nobs"], seed=7)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="gpu", diff=args.diff, schur=args.schur)
    print("forward+chi2        %7.2f ms" % t(p.forward))
    p.accept_forward()
    print("assembly(+transp)   %7.2f ms" % t(p.build_linear_system))
    print("process_diag        %7.2f ms" % t(lambda: p.process_diag(1e4)))
    iters = [0]

    def solve():
        iters[0] = p.solve_linear(max_iter=100, tol=1e-1, refuse_ratio=1.0)
    ms = t(solve, reps=3)
    print("solve_linear        %7.2f ms  (%d PCG iters -> %.0f us/iter)"
          % (ms, iters[0], 1000 * ms / max(iters[0], 1)))
    print("delta_x_l2          %7.2f ms" % t(p.delta_x_l2))
    print("x_l2                %7.2f ms" % t(p.x_l2))
    print("g_inf               %7.2f ms" % t(p.g_inf))
    print("update_params       %7.2f ms" % t(p.update_params))
    print("rho_denominator     %7.2f ms" % t(lambda: p.rho_denominator(0.0)))
    sys.stdout.flush()


if __name__ == "__main__":
    main()
