import numpy as np
import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an AMD GPU (run with -m gpu on an MI355X box)")


def pytest_collection_modifyitems(config, items):
    # Skip gpu tests automatically when no GPU is present, unless -m gpu was
    # requested explicitly (then a missing GPU should fail loudly).
    markexpr = config.getoption("-m") or ""
    if "gpu" in markexpr:
        return
    try:
        from megba_amd import _core
        ngpu = _core.hip_device_count() if _core.has_gpu_support else 0
    except Exception:
        ngpu = 0
    if ngpu == 0:
        skip = pytest.mark.skip(reason="no GPU available")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)


@pytest.fixture
def small_problem():
    import megba_amd as mb
    cams, pts, ci, pi, meas = mb.synthesize_bal(12, 120, 1100, seed=3)
    return cams, pts, ci, pi, meas


def bal_residual_np(cam, pt, meas):
    """Independent numpy implementation of the BAL residual (oracle)."""
    aa = cam[:3]
    theta = np.linalg.norm(aa)
    x = pt
    if theta > 1e-7:
        w = aa / theta
        c, s = np.cos(theta), np.sin(theta)
        P = x * c + np.cross(w, x) * s + w * np.dot(w, x) * (1 - c)
    else:
        P = x + np.cross(aa, x)
    P = P + cam[3:6]
    p = -P[:2] / P[2]
    r2 = p @ p
    dist = 1.0 + cam[7] * r2 + cam[8] * r2 * r2
    return cam[6] * dist * p - meas


def dense_reference(dump, ncam, npt, info=None):
    """Build the dense damped normal equations from a dumped J set."""
    nL = int(dump["e1"] - dump["e0"])
    Jc = dump["Jc"].reshape(nL, 2, 9)
    Jp = dump["Jp"].reshape(nL, 2, 3)
    r = dump["r"].reshape(nL, 2)
    return nL, Jc, Jp, r
