import megba_amd as mb
cams,pts,ci,pi,meas=mb.synthesize_bal(1778,993923,5000000,seed=7)
p=mb.BAProblem(cams,pts,ci,pi,meas)
p.build(device="gpu", schur="implicit")
p.lm_init(tau=1e4, solver_max_iter=8, solver_tol=0.0, solver_refuse_ratio=1e30, force_iterations=True, verbose=False)
for _ in range(2): p.lm_step()
print("pmc workload done")
