"""Exhaustive mode matrix: every build-option combination must solve a
small problem to a finite, decreasing cost.  96 combos (dtype x diff x
schur x loss x info x fixed) on CPU — the GPU equivalents of each axis
are covered pairwise in the other suites."""
import itertools

import numpy as np
import pytest

import megba_amd as mb

CAMS, PTS, CI, PI, MEAS = mb.synthesize_bal(6, 40, 280, seed=17)
RNG = np.random.default_rng(0)
INFO = np.zeros((len(CI), 3))
INFO[:, 0] = RNG.uniform(0.5, 2.0, len(CI))
INFO[:, 2] = RNG.uniform(0.5, 2.0, len(CI))
CAM_FIXED = np.zeros(len(CAMS), dtype=np.uint8)
CAM_FIXED[0] = 1

MATRIX = list(itertools.product(
    ["float64", "float32"],
    ["auto", "analytical"],
    ["explicit", "implicit"],
    ["none", "huber", "cauchy"],
    [False, True],   # info
    [False, True],   # fixed camera 0
))


@pytest.mark.parametrize("dtype,diff,schur,loss,with_info,with_fixed",
                         MATRIX)
def test_combo(dtype, diff, schur, loss, with_info, with_fixed):
    p = mb.BAProblem(CAMS, PTS, CI, PI, MEAS,
                     info=INFO if with_info else None,
                     cam_fixed=CAM_FIXED if with_fixed else None)
    p.build(device="cpu", dtype=dtype, diff=diff, schur=schur, loss=loss,
            loss_delta=2.0)
    rep = p.solve(max_iter=3, tau=1e4, solver_tol=1e-6, solver_max_iter=80,
                  solver_refuse_ratio=1e9, verbose=False)
    assert np.isfinite(rep["final_chi2"])
    assert rep["final_chi2"] < rep["iters"][0]["chi2"]
    if with_fixed:
        c, _ = p.get_params()
        # fp32 engines store parameters in float32: the fixed vertex is
        # unchanged up to the storage cast.
        want = CAMS[0].astype(np.float32) if dtype == "float32" else CAMS[0]
        np.testing.assert_array_equal(c[0].astype(want.dtype), want)
