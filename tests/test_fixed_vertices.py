"""Fixed-vertex support (g2o parity): fixed cameras/points must not move and
the rest must still optimise."""
import numpy as np
import pytest

import megba_amd as mb


def _mk(device):
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    cam_fixed = np.zeros(12, dtype=np.uint8)
    cam_fixed[:3] = 1
    pt_fixed = np.zeros(120, dtype=np.uint8)
    pt_fixed[::10] = 1
    p = mb.BAProblem(cams, pts, ci, pi, meas, cam_fixed=cam_fixed,
                     pt_fixed=pt_fixed)
    p.build(device=device)
    return p, cams, pts, cam_fixed, pt_fixed


def _check(p, cams, pts, cam_fixed, pt_fixed):
    rep = p.solve(max_iter=6, verbose=False)
    assert rep["final_chi2"] < rep["iters"][0]["chi2"]
    c2, p2 = p.get_params()
    np.testing.assert_array_equal(c2[cam_fixed == 1], cams[cam_fixed == 1])
    np.testing.assert_array_equal(p2[pt_fixed == 1], pts[pt_fixed == 1])
    assert not np.allclose(c2[cam_fixed == 0], cams[cam_fixed == 0])


def test_fixed_cpu():
    _check(*_mk("cpu"))


@pytest.mark.gpu
def test_fixed_gpu():
    _check(*_mk("gpu"))


@pytest.mark.gpu
def test_fixed_gpu_matches_cpu():
    pc, *_ = _mk("cpu")
    pg, *_ = _mk("gpu")
    kw = dict(max_iter=6, solver_tol=1e-6, solver_max_iter=300,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = pc.solve(**kw), pg.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-5)
