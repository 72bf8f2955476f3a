"""Solver-knob semantics parity (reference SolverOptionPCG,
common.h:27-34 + schur_pcg_solver.cu:266-295): absolute |r^T z| < tol
convergence, refuseRatio early-stop with x restore, warm start from the
previous deltaX."""
import numpy as np

import megba_amd as mb


def _ready_problem(seed=5):
    cams, pts, ci, pi, meas = mb.synthesize_bal(8, 60, 460, seed=seed)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu")
    p.forward()
    p.accept_forward()
    p.build_linear_system()
    p.process_diag(1e4)
    return p


def test_tol_monotone_iteration_count():
    # Looser absolute tolerance must not need more PCG iterations.
    counts = []
    for tol in (1e-12, 1e-6, 1e0, 1e6):
        p = _ready_problem()
        counts.append(p.solve_linear(max_iter=500, tol=tol,
                                     refuse_ratio=1e18))
    assert counts == sorted(counts, reverse=True), counts
    assert counts[0] > counts[-1]  # the sweep actually spans regimes


def test_max_iter_caps_work():
    p = _ready_problem()
    assert p.solve_linear(max_iter=7, tol=0.0, refuse_ratio=1e30) == 7


def test_refuse_ratio_early_stop():
    # refuse=1.0 stops as soon as rho exceeds its running minimum
    # (reference semantics); a huge ratio runs to tol.
    p1 = _ready_problem()
    it_strict = p1.solve_linear(max_iter=500, tol=1e-12, refuse_ratio=1.0)
    p2 = _ready_problem()
    it_loose = p2.solve_linear(max_iter=500, tol=1e-12, refuse_ratio=1e18)
    assert it_strict <= it_loose
    # and the refused solve still leaves a usable (finite) deltaX
    d = p1.dump()
    assert np.isfinite(d["deltaX"]).all()


def test_warm_start_reuses_previous_solution():
    # Second solve of the SAME system warm-starts from the previous deltaX
    # and must converge in (far) fewer iterations.
    p = _ready_problem()
    first = p.solve_linear(max_iter=500, tol=1e-8, refuse_ratio=1e18)
    second = p.solve_linear(max_iter=500, tol=1e-8, refuse_ratio=1e18)
    assert second < first, (first, second)
