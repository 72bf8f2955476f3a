# Small venice fp64 fixed-work workload for PMC on the product kernels.
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import megba_amd as mb
cams, pts, ci, pi, meas = mb.synthesize_bal(1778, 993923, 5000000, seed=7)
p = mb.BAProblem(cams, pts, ci, pi, meas)
p.build(device="gpu", schur="implicit")
p.lm_init(tau=1e4, solver_max_iter=6, solver_tol=0.0,
          solver_refuse_ratio=1e30, force_iterations=True, verbose=False)
for _ in range(2):
    p.lm_step()
print("done")
