import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
import megba_amd as mb
from megba_amd import _core
cams9, pts, ci, pi, meas = mb.synthesize_bal(1778, 993923, 5_000_000, seed=7)
cams6 = np.ascontiguousarray(cams9[:, :6])
intr = [1000.0, 0.0, 0.0]
cams9[:, 6] = intr[0]; cams9[:, 7] = intr[1]; cams9[:, 8] = intr[2]
p = mb.BAProblem(cams6, pts, ci, pi, meas)
p.build(device="gpu", schur="implicit", intrinsics=intr)
p.lm_init(tau=1e4, solver_max_iter=100, solver_tol=0.0,
          solver_refuse_ratio=1e30, force_iterations=True, verbose=False)
for _ in range(3): p.lm_step()
_core.device_synchronize()
t0 = time.perf_counter()
for _ in range(10): p.lm_step()
_core.device_synchronize()
print(f"(6,3,2) venice-shape fp64 implicit: {(time.perf_counter()-t0)*100:.2f} ms/step")
