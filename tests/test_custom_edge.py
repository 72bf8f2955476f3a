"""Runtime user-defined edges: the BAL reprojection residual expressed as a
Python forward() over JetVectors must reproduce the built-in fused path
(r, J, and the whole LM trajectory)."""
import numpy as np
import pytest

import megba_amd as mb
from megba_amd import jv


def bal_forward(cam, pt, meas):
    cam, pt, meas = jv.wrap(cam), jv.wrap(pt), jv.wrap(meas)
    R = jv.angle_axis_to_rotation(cam[0:3])
    P = [R[3 * i] * pt[0] + R[3 * i + 1] * pt[1] + R[3 * i + 2] * pt[2]
         + cam[3 + i] for i in range(3)]
    px = -P[0] / P[2]
    py = -P[1] / P[2]
    fr = jv.radial_distortion([px, py], cam[6:9])
    return ((fr * px - meas[0]).raw, (fr * py - meas[1]).raw)


def _pair(device):
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    ref = mb.BAProblem(cams, pts, ci, pi, meas)
    ref.build(device=device)
    cust = mb.BAProblem(cams, pts, ci, pi, meas)
    cust.build(device=device, custom_forward=bal_forward)
    return ref, cust


def _check_forward(ref, cust):
    c1 = ref.forward()
    c2 = cust.forward()
    np.testing.assert_allclose(c2, c1, rtol=1e-9)
    d1, d2 = ref.dump(), cust.dump()
    for key in ("r", "Jc", "Jp"):
        scale = np.abs(d1[key]).max()
        np.testing.assert_allclose(d2[key], d1[key], rtol=1e-8,
                                   atol=1e-9 * scale, err_msg=key)


def test_custom_forward_cpu():
    _check_forward(*_pair("cpu"))


def test_custom_solve_cpu():
    ref, cust = _pair("cpu")
    kw = dict(max_iter=5, solver_tol=1e-6, solver_max_iter=200,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ref.solve(**kw), cust.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-6)


@pytest.mark.gpu
def test_custom_forward_gpu():
    _check_forward(*_pair("gpu"))


@pytest.mark.gpu
def test_custom_solve_gpu():
    ref, cust = _pair("gpu")
    kw = dict(max_iter=5, solver_tol=1e-6, solver_max_iter=200,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ref.solve(**kw), cust.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-6)


def _pair_loss(device):
    # custom forward composed with robust loss: the engine's chi2 must apply
    # rho to the user-residual (kChi2Loss on GPU), and assembly must weight
    # the repacked J rows identically to the built-in path.
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=13)
    rng = np.random.default_rng(1)
    bad = rng.random(len(ci)) < 0.1
    meas = meas.copy()
    meas[bad] += rng.normal(scale=40.0, size=(int(bad.sum()), 2))
    ref = mb.BAProblem(cams, pts, ci, pi, meas)
    ref.build(device=device, loss="huber", loss_delta=2.0)
    cust = mb.BAProblem(cams, pts, ci, pi, meas)
    cust.build(device=device, loss="huber", loss_delta=2.0,
               custom_forward=bal_forward)
    return ref, cust


def test_custom_forward_with_loss_cpu():
    ref, cust = _pair_loss("cpu")
    kw = dict(max_iter=5, solver_tol=1e-6, solver_max_iter=200,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ref.solve(**kw), cust.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-6)


@pytest.mark.gpu
def test_custom_forward_with_loss_gpu():
    ref, cust = _pair_loss("gpu")
    kw = dict(max_iter=5, solver_tol=1e-6, solver_max_iter=200,
              solver_refuse_ratio=1e6, verbose=False)
    r1, r2 = ref.solve(**kw), cust.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-6)


def _run_cpp_custom(device, tmp_path):
    """The native C++ custom-edge example (user forward() in C++ over the
    JetVector op layer, reference examples/BAL_Double.cpp:16-34) is
    self-verifying: it solves the same problem through the built-in fused
    path and the custom callback and compares trajectories."""
    import os
    import subprocess
    binpath = "examples/bal_custom_edge_cpp"
    if not os.path.exists(binpath):
        pytest.skip("native example not built")
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    f = tmp_path / "prob.txt"
    mb.save_bal(f, cams, pts, ci, pi, meas)
    r = subprocess.run([binpath, "--path", str(f), "--device", device,
                        "--max_iter", "5"],
                       capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:] + r.stdout[-2000:]
    assert "CUSTOM_MATCH_OK" in r.stdout


def test_cpp_custom_edge_cpu(tmp_path):
    _run_cpp_custom("cpu", tmp_path)


@pytest.mark.gpu
def test_cpp_custom_edge_gpu(tmp_path):
    _run_cpp_custom("gpu", tmp_path)
