"""CPU-path numerics: Jacobians vs finite differences, assembly vs dense
J^T W J, PCG solution vs dense linear solve, LM convergence."""
import numpy as np
import pytest

import megba_amd as mb
from conftest import bal_residual_np


def build_cpu(small_problem, **kw):
    cams, pts, ci, pi, meas = small_problem
    p = mb.BAProblem(cams, pts, ci, pi, meas, **kw)
    p.build(device="cpu")
    return p


def test_forward_matches_numpy(small_problem):
    cams, pts, ci, pi, meas = small_problem
    p = build_cpu(small_problem)
    chi2 = p.forward()
    d = p.dump()
    ii = p.index_info()
    nL = len(ci)
    r = d["r"].reshape(nL, 2)
    # residuals against the independent numpy model (sorted edge order)
    cam_of, pt_of = ii["cam_of"], ii["pt_of"]
    perm = ii["perm"]
    chk = 0.0
    for e in range(0, nL, 7):
        rr = bal_residual_np(cams[cam_of[e]], pts[pt_of[e]],
                             meas[perm[e]])
        np.testing.assert_allclose(r[e], rr, rtol=1e-10, atol=1e-12)
    np.testing.assert_allclose(chi2, (d["r"] ** 2).sum(), rtol=1e-12)


def test_jacobian_matches_finite_difference(small_problem):
    cams, pts, ci, pi, meas = small_problem
    p = build_cpu(small_problem)
    p.forward()
    d = p.dump()
    ii = p.index_info()
    nL = len(ci)
    Jc = d["Jc"].reshape(nL, 2, 9)
    Jp = d["Jp"].reshape(nL, 2, 3)
    rng = np.random.default_rng(0)
    eps = 1e-7
    for e in rng.integers(0, nL, size=25):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        m = meas[ii["perm"][e]]
        for k in range(9):
            cp = cams[c].copy()
            cm = cams[c].copy()
            cp[k] += eps
            cm[k] -= eps
            fd = (bal_residual_np(cp, pts[q], m) -
                  bal_residual_np(cm, pts[q], m)) / (2 * eps)
            scale = max(1.0, np.abs(fd).max())
            np.testing.assert_allclose(Jc[e, :, k], fd, atol=2e-4 * scale)
        for k in range(3):
            pp = pts[q].copy()
            pm = pts[q].copy()
            pp[k] += eps
            pm[k] -= eps
            fd = (bal_residual_np(cams[c], pp, m) -
                  bal_residual_np(cams[c], pm, m)) / (2 * eps)
            scale = max(1.0, np.abs(fd).max())
            np.testing.assert_allclose(Jp[e, :, k], fd, atol=2e-4 * scale)


def _dense_system(p, ncam, npt, info=None, d=None):
    if d is None:
        d = p.dump()
    nL = int(d["e1"] - d["e0"])
    Jc = d["Jc"].reshape(nL, 2, 9)
    Jp = d["Jp"].reshape(nL, 2, 3)
    r = d["r"].reshape(nL, 2)
    ii = p.index_info()
    dim = 9 * ncam + 3 * npt
    J = np.zeros((2 * nL, dim))
    for e in range(nL):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        J[2 * e:2 * e + 2, 9 * c:9 * c + 9] = Jc[e]
        J[2 * e:2 * e + 2, 9 * ncam + 3 * q:9 * ncam + 3 * q + 3] = Jp[e]
    rv = r.reshape(-1)
    if info is not None:
        W = np.zeros((2 * nL, 2 * nL))
        for e in range(nL):
            W[2 * e:2 * e + 2, 2 * e:2 * e + 2] = [
                [info[e, 0], info[e, 1]], [info[e, 1], info[e, 2]]]
        H = J.T @ W @ J
        g = -J.T @ (W @ rv)
    else:
        H = J.T @ J
        g = -J.T @ rv
    return H, g, d


def test_assembly_matches_dense(small_problem):
    cams, pts, ci, pi, meas = small_problem
    ncam, npt = len(cams), len(pts)
    p = build_cpu(small_problem)
    p.forward()
    dj = p.dump()          # J/r of the last forward (before the accept swap)
    p.accept_forward()
    p.build_linear_system()
    H, g, _ = _dense_system(p, ncam, npt, d=dj)
    d = p.dump()
    Hpp = d["Hpp"].reshape(ncam, 9, 9)
    Hll = d["Hll"].reshape(npt, 3, 3)
    for c in range(ncam):
        np.testing.assert_allclose(Hpp[c], H[9 * c:9 * c + 9, 9 * c:9 * c + 9],
                                   rtol=1e-8, atol=1e-8)
    for q in range(0, npt, 5):
        o = 9 * ncam + 3 * q
        np.testing.assert_allclose(Hll[q], H[o:o + 3, o:o + 3],
                                   rtol=1e-8, atol=1e-8)
    np.testing.assert_allclose(d["g"], g, rtol=1e-8, atol=1e-8)
    # Hpl blocks
    ii = p.index_info()
    nL = len(ci)
    Hpl = d["Hpl"].reshape(nL, 9, 3)
    for e in range(0, nL, 11):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        blk = H[9 * c:9 * c + 9, 9 * ncam + 3 * q:9 * ncam + 3 * q + 3]
        # multiple edges can share a (cam,pt) pair only if duplicated; synth
        # may duplicate, so compare the SUM over matching edges.
        mask = (ii["cam_of"] == c) & (ii["pt_of"] == q)
        total = Hpl[mask].sum(axis=0)
        np.testing.assert_allclose(total, blk, rtol=1e-8, atol=1e-8)


def test_weighted_assembly(small_problem):
    cams, pts, ci, pi, meas = small_problem
    ncam, npt = len(cams), len(pts)
    rng = np.random.default_rng(5)
    # SPD 2x2 information per edge (in ORIGINAL edge order)
    a = rng.uniform(0.5, 2.0, size=len(ci))
    b = rng.uniform(0.5, 2.0, size=len(ci))
    c01 = rng.uniform(-0.3, 0.3, size=len(ci)) * np.sqrt(a * b)
    info = np.stack([a, c01, b], axis=1)
    p = mb.BAProblem(cams, pts, ci, pi, meas, info=info)
    p.build(device="cpu")
    p.forward()
    dj = p.dump()
    p.accept_forward()
    p.build_linear_system()
    ii = p.index_info()
    info_sorted = info[ii["perm"]]
    H, g, _ = _dense_system(p, ncam, npt, info=info_sorted, d=dj)
    d = p.dump()
    np.testing.assert_allclose(d["g"], g, rtol=1e-8, atol=1e-8)
    Hpp = d["Hpp"].reshape(ncam, 9, 9)
    for c in range(0, ncam, 3):
        np.testing.assert_allclose(Hpp[c], H[9 * c:9 * c + 9, 9 * c:9 * c + 9],
                                   rtol=1e-8, atol=1e-8)


def test_pcg_matches_dense_solve(small_problem):
    cams, pts, ci, pi, meas = small_problem
    ncam, npt = len(cams), len(pts)
    p = build_cpu(small_problem)
    p.forward()
    dj = p.dump()
    p.accept_forward()
    p.build_linear_system()
    region = 1e4
    p.process_diag(region)
    iters = p.solve_linear(max_iter=2000, tol=1e-16, refuse_ratio=1e18)
    d = p.dump()
    H, g, _ = _dense_system(p, ncam, npt, d=dj)
    f = 1.0 + 1.0 / region
    Hd = H.copy()
    idx = np.arange(H.shape[0])
    Hd[idx, idx] *= f
    dx_ref = np.linalg.solve(Hd, g)
    scale = np.abs(dx_ref).max()
    np.testing.assert_allclose(d["deltaX"], dx_ref, atol=1e-6 * scale)
    assert iters > 0


def test_lm_decreases_chi2(small_problem):
    p = build_cpu(small_problem)
    rep = p.solve(max_iter=20, tau=1e4, solver_tol=1e-3, solver_max_iter=200,
                  solver_refuse_ratio=100.0, verbose=False)
    chis = [it["chi2"] for it in rep["iters"] if it["accepted"]]
    assert all(b <= a * (1 + 1e-12) for a, b in zip(chis, chis[1:]))
    assert rep["final_chi2"] < 0.2 * rep["iters"][0]["chi2"]


def test_fp32_engine_runs(small_problem):
    cams, pts, ci, pi, meas = small_problem
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu", dtype="float32")
    rep = p.solve(max_iter=6, verbose=False)
    assert rep["final_chi2"] < rep["iters"][0]["chi2"]
