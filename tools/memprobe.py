# Device-memory footprint probe: build + one forward, then hipMemGetInfo.
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import megba_amd as mb
from megba_amd import _core
model, dtype, schur, diff = sys.argv[1:5]
shapes = {"venice1778": (1778, 993923, 5000000),
          "final13682": (13682, 4456117, 28987644),
          "synth20k": (20000, 10000000, 50000000)}
ncam, npt, nobs = shapes[model]
cams, pts, ci, pi, meas = mb.synthesize_bal(ncam, npt, nobs, seed=7)
free0, total = _core.hip_mem_info()
p = mb.BAProblem(cams, pts, ci, pi, meas)
p.build(device="gpu", dtype=dtype, schur=schur, diff=diff)
p.forward()
free1, _ = _core.hip_mem_info()
print(f"{model} {dtype} {schur}: {(free0-free1)/2**30:.1f} GiB")
