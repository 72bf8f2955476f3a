"""LM stop criteria (reference semantics, lm_algo.cu:139-223 / SURVEY C19):
epsilon1 bounds the gradient inf-norm after an accepted step; epsilon2
stops when ||dx|| <= eps2 * (||x|| + eps1); force_iterations disables both
(the benchmark mode)."""
import megba_amd as mb


def _problem():
    cams, pts, ci, pi, meas = mb.synthesize_bal(6, 50, 380, seed=9)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu")
    return p


def _solve(p, **kw):
    base = dict(max_iter=20, tau=1e4, epsilon1=1e-12, epsilon2=1e-14,
                solver_tol=1e-8, solver_max_iter=200,
                solver_refuse_ratio=1e9, verbose=False)
    base.update(kw)
    return p.solve(**base)


def test_epsilon1_gradient_stop():
    # Gigantic epsilon1: any accepted step satisfies ||g||_inf <= eps1 and
    # the loop must stop far before max_iter.
    rep = _solve(_problem(), epsilon1=1e12)
    assert len(rep["iters"]) - 1 < 20


def test_epsilon2_step_size_stop():
    # Gigantic epsilon2: the ||dx|| criterion fires on the first check.
    rep = _solve(_problem(), epsilon2=1.0)
    assert len(rep["iters"]) - 1 <= 2


def test_force_iterations_ignores_criteria():
    p = _problem()
    rep = p.solve(max_iter=7, epsilon1=1e12, epsilon2=1.0,
                  force_iterations=True, verbose=False)
    assert len(rep["iters"]) - 1 == 7


def test_default_run_converges_before_cap():
    # Noise-free measurements: the optimum is (near) zero residual, so a
    # converged run must reduce chi2 by orders of magnitude.
    cams, pts, ci, pi, meas = mb.synthesize_bal(6, 50, 380, seed=9,
                                                pixel_noise=0.0)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu")
    rep = _solve(p, epsilon1=1e-8)
    assert rep["final_chi2"] < 1e-3 * rep["iters"][0]["chi2"]
