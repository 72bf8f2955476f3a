"""Pathological sparsity structures: a hub point seen by every camera, many
degree-1 points (numerically singular Hll blocks -> jitter-invert path),
uneven camera degrees.  CPU and GPU must agree."""
import numpy as np
import pytest

import megba_amd as mb


def _mk():
    rng = np.random.default_rng(7)
    ncam, npt = 6, 30
    cams, pts, _, _, _ = mb.synthesize_bal(ncam, npt, 200, seed=7)
    ci, pi = [], []
    # hub point 0: seen by every camera
    for c in range(ncam):
        ci.append(c)
        pi.append(0)
    # degree-1 points (singular 3x3 blocks)
    for q in range(1, 10):
        ci.append(int(rng.integers(0, ncam)))
        pi.append(q)
    # the rest: degree 2+
    for q in range(10, npt):
        for c in rng.choice(ncam, size=2, replace=False):
            ci.append(int(c))
            pi.append(q)
    # camera degree floor
    for c in range(ncam):
        ci.append(c)
        pi.append(int(rng.integers(10, npt)))
    ci = np.array(ci, dtype=np.int32)
    pi = np.array(pi, dtype=np.int32)
    meas = rng.normal(scale=5.0, size=(len(ci), 2))  # arbitrary targets
    return cams, pts, ci, pi, meas


def _state(device):
    cams, pts, ci, pi, meas = _mk()
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device=device)
    p.forward()
    p.accept_forward()
    p.build_linear_system()
    p.process_diag(1e4)
    p.solve_linear(max_iter=300, tol=1e-12, refuse_ratio=1e18)
    return p.dump()


def test_cpu_runs_skewed():
    d = _state("cpu")
    assert np.isfinite(d["deltaX"]).all()


@pytest.mark.gpu
def test_gpu_matches_cpu_skewed():
    d1 = _state("cpu")
    d2 = _state("gpu")
    for key in ("Hpp", "Hll", "g"):
        scale = np.abs(d1[key]).max() or 1.0
        np.testing.assert_allclose(d2[key], d1[key], rtol=1e-8,
                                   atol=1e-9 * scale, err_msg=key)
    scale = np.abs(d1["deltaX"]).max()
    np.testing.assert_allclose(d2["deltaX"], d1["deltaX"], atol=5e-4 * scale)
