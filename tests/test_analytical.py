"""Analytical (closed-form) Jacobians must match the autodiff (Jet) path."""
import numpy as np
import pytest

import megba_amd as mb


def _pair(device, shape=(12, 120, 1100), seed=3):
    cams, pts, ci, pi, meas = mb.synthesize_bal(*shape, seed=seed)
    auto = mb.BAProblem(cams, pts, ci, pi, meas)
    auto.build(device=device, diff="auto")
    ana = mb.BAProblem(cams, pts, ci, pi, meas)
    ana.build(device=device, diff="analytical")
    return auto, ana


def check(auto, ana):
    c1 = auto.forward()
    c2 = ana.forward()
    np.testing.assert_allclose(c2, c1, rtol=1e-10)
    d1, d2 = auto.dump(), ana.dump()
    for key in ("r", "Jc", "Jp"):
        scale = np.abs(d1[key]).max()
        np.testing.assert_allclose(d2[key], d1[key], rtol=1e-7,
                                   atol=1e-9 * scale, err_msg=key)


def test_analytical_matches_autodiff_cpu():
    auto, ana = _pair("cpu")
    check(auto, ana)


def test_analytical_small_angle_cpu():
    # Force the theta ~ 0 branch.
    cams, pts, ci, pi, meas = mb.synthesize_bal(6, 60, 500, seed=2)
    cams[:, :3] = np.random.default_rng(0).normal(scale=1e-9, size=(6, 3))
    auto = mb.BAProblem(cams, pts, ci, pi, meas)
    auto.build(device="cpu", diff="auto")
    ana = mb.BAProblem(cams, pts, ci, pi, meas)
    ana.build(device="cpu", diff="analytical")
    check(auto, ana)


def test_analytical_solve_cpu():
    auto, ana = _pair("cpu")
    kw = dict(max_iter=6, solver_tol=1e-6, solver_max_iter=200,
              solver_refuse_ratio=1e6, verbose=False)
    r1 = auto.solve(**kw)
    r2 = ana.solve(**kw)
    c1 = [it["chi2"] for it in r1["iters"]]
    c2 = [it["chi2"] for it in r2["iters"]]
    np.testing.assert_allclose(c2, c1, rtol=1e-5)


@pytest.mark.gpu
def test_analytical_matches_autodiff_gpu():
    auto, ana = _pair("gpu")
    check(auto, ana)


@pytest.mark.gpu
def test_analytical_gpu_matches_cpu():
    cpu_auto, cpu_ana = _pair("cpu")
    gpu_auto, gpu_ana = _pair("gpu")
    cpu_ana.forward()
    gpu_ana.forward()
    d1, d2 = cpu_ana.dump(), gpu_ana.dump()
    for key in ("r", "Jc", "Jp"):
        scale = np.abs(d1[key]).max()
        np.testing.assert_allclose(d2[key], d1[key], rtol=1e-9,
                                   atol=1e-10 * scale, err_msg=key)
