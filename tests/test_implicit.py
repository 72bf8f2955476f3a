"""Implicit Schur (matrix-free E/E^T from J) must match the explicit path."""
import numpy as np
import pytest

import megba_amd as mb


def _pair(device, info=None):
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    ex = mb.BAProblem(cams, pts, ci, pi, meas, info=info)
    ex.build(device=device, schur="explicit")
    im = mb.BAProblem(cams, pts, ci, pi, meas, info=info)
    im.build(device=device, schur="implicit")
    return ex, im


def _solve_dx(p):
    p.forward()
    p.accept_forward()
    p.build_linear_system()
    p.process_diag(1e4)
    p.solve_linear(max_iter=500, tol=1e-14, refuse_ratio=1e18)
    return p.dump()["deltaX"]


def test_implicit_matches_explicit_cpu():
    ex, im = _pair("cpu")
    d1, d2 = _solve_dx(ex), _solve_dx(im)
    scale = np.abs(d1).max()
    np.testing.assert_allclose(d2, d1, atol=1e-7 * scale)


def test_implicit_weighted_cpu():
    rng = np.random.default_rng(5)
    n = 1100
    a = rng.uniform(0.5, 2.0, size=n)
    b = rng.uniform(0.5, 2.0, size=n)
    c01 = rng.uniform(-0.3, 0.3, size=n) * np.sqrt(a * b)
    info = np.stack([a, c01, b], axis=1)
    ex, im = _pair("cpu", info=info)
    d1, d2 = _solve_dx(ex), _solve_dx(im)
    scale = np.abs(d1).max()
    np.testing.assert_allclose(d2, d1, atol=1e-7 * scale)


def test_implicit_full_solve_cpu():
    ex, im = _pair("cpu")
    kw = dict(max_iter=6, solver_tol=1e-6, solver_max_iter=300,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ex.solve(**kw), im.solve(**kw)
    c1 = [it["chi2"] for it in r1["iters"]]
    c2 = [it["chi2"] for it in r2["iters"]]
    np.testing.assert_allclose(c2, c1, rtol=1e-5)


@pytest.mark.gpu
def test_implicit_matches_explicit_gpu():
    ex, im = _pair("gpu")
    d1, d2 = _solve_dx(ex), _solve_dx(im)
    scale = np.abs(d1).max()
    np.testing.assert_allclose(d2, d1, atol=1e-6 * scale)


@pytest.mark.gpu
def test_implicit_fp32_analytical_gpu():
    # BASELINE config 4 combination: fp32 + analytical + implicit.
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="gpu", dtype="float32", diff="analytical", schur="implicit")
    rep = p.solve(max_iter=5, verbose=False)
    assert rep["final_chi2"] < rep["iters"][0]["chi2"]


@pytest.mark.gpu
def test_implicit_full_solve_gpu():
    # Multiple accept/reject cycles: catches stale-buffer bugs in the
    # implicit path (the PCG operator must always read the freshly accepted
    # Jacobians, even across the double-buffer swaps).
    ex, im = _pair("gpu")
    kw = dict(max_iter=8, solver_tol=1e-6, solver_max_iter=300,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ex.solve(**kw), im.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-5)
