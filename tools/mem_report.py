#!/usr/bin/env python3
"""Report device-memory footprint of a built problem (run via gpurun)."""
import argparse
import sys

import megba_amd as mb
from megba_amd import _core


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="venice1778")
    ap.add_argument("--dtype", default="float64")
    ap.add_argument("--schur", default="explicit")
    args = ap.parse_args()
    from bench import MODELS
    shape = MODELS[args.model]
    free0, total = _core.hip_mem_info()
    cams, pts, ci, pi, meas = mb.synthesize_bal(
        shape["ncam"], shape["npt"], shape["nobs"], seed=7)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="gpu", dtype=args.dtype, schur=args.schur)
    p.forward()
    free1, _ = _core.hip_mem_info()
    gb = (free0 - free1) / 2**30
    print(f"{args.model} {args.dtype} {args.schur}: {gb:.1f} GiB device memory "
          f"({(total-free1)/2**30:.1f} of {total/2**30:.0f} GiB in use)")
    sys.stdout.flush()


if __name__ == "__main__":
    main()
