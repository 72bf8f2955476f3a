# Small fixed-work final13682-fp32 workload for PMC collection on the two
# packed implicit product kernels (few PCG iterations to bound counter time).
import megba_amd as mb
cams, pts, ci, pi, meas = mb.synthesize_bal(13682, 4456117, 28987644, seed=7)
p = mb.BAProblem(cams, pts, ci, pi, meas)
p.build(device="gpu", dtype="float32", diff="analytical", schur="implicit")
p.lm_init(tau=1e4, solver_max_iter=6, solver_tol=0.0,
          solver_refuse_ratio=1e30, force_iterations=True, verbose=False)
for _ in range(2):
    p.lm_step()
print("pmc workload done")
