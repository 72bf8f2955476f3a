"""A quaternion-parameterised camera model via the runtime custom-edge path:
cam = [q(4), t(3), f, k1].  Demonstrates that user-defined models with the
JetVector geometry ops drive the full LM+Schur solver."""
import numpy as np
import pytest

import megba_amd as mb
from megba_amd import jv


def quat_forward(cam, pt, meas):
    cam, pt, meas = jv.wrap(cam), jv.wrap(pt), jv.wrap(meas)
    q = jv.normalize_quaternion(cam[0:4])
    R = jv.quaternion_to_rotation(q)
    P = [R[3 * i] * pt[0] + R[3 * i + 1] * pt[1] + R[3 * i + 2] * pt[2]
         + cam[4 + i] for i in range(3)]
    px = -P[0] / P[2]
    py = -P[1] / P[2]
    r2 = px * px + py * py
    fr = cam[7] * (1.0 + cam[8] * r2)
    return ((fr * px - meas[0]).raw, (fr * py - meas[1]).raw)


def _aa_to_quat(aa):
    th = np.linalg.norm(aa, axis=1, keepdims=True)
    w = np.cos(th / 2)
    xyz = np.where(th > 1e-12, np.sin(th / 2) * aa / np.maximum(th, 1e-12),
                   aa / 2)
    return np.concatenate([w, xyz], axis=1)


def _make(seed=3):
    rng = np.random.default_rng(seed)
    cams_aa, pts, ci, pi, _ = mb.synthesize_bal(10, 100, 900, seed=seed,
                                                pixel_noise=0.0)
    # quaternion camera: [q(4), t(3), f, k1]
    cams = np.zeros((len(cams_aa), 9))
    cams[:, 0:4] = _aa_to_quat(cams_aa[:, 0:3])
    cams[:, 4:7] = cams_aa[:, 3:6]
    cams[:, 7] = cams_aa[:, 6]
    cams[:, 8] = cams_aa[:, 7]
    # ground-truth measurements from the quaternion model itself
    def project(c, x):
        w, qx, qy, qz = c[0:4] / np.linalg.norm(c[0:4])
        R = np.array([
            [1 - 2 * (qy * qy + qz * qz), 2 * (qx * qy - w * qz), 2 * (qx * qz + w * qy)],
            [2 * (qx * qy + w * qz), 1 - 2 * (qx * qx + qz * qz), 2 * (qy * qz - w * qx)],
            [2 * (qx * qz - w * qy), 2 * (qy * qz + w * qx), 1 - 2 * (qx * qx + qy * qy)]])
        P = R @ x + c[4:7]
        p = -P[:2] / P[2]
        return c[7] * (1 + c[8] * (p @ p)) * p
    meas = np.array([project(cams[ci[e]], pts[pi[e]]) for e in range(len(ci))])
    # perturb the initial estimate
    cams0 = cams.copy()
    cams0[:, 0:4] += rng.normal(scale=2e-3, size=(len(cams), 4))
    cams0[:, 4:7] += rng.normal(scale=1e-2, size=(len(cams), 3))
    pts0 = pts + rng.normal(scale=1e-2, size=pts.shape)
    return cams0, pts0, ci, pi, meas


def _run(device):
    cams0, pts0, ci, pi, meas = _make()
    p = mb.BAProblem(cams0, pts0, ci, pi, meas)
    p.build(device=device, custom_forward=quat_forward)
    rep = p.solve(max_iter=10, tau=1e4, solver_tol=1e-6, solver_max_iter=200,
                  solver_refuse_ratio=1e6, verbose=False)
    assert rep["final_chi2"] < 1e-2 * rep["iters"][0]["chi2"], rep["iters"]


def test_quaternion_edge_cpu():
    _run("cpu")


@pytest.mark.gpu
def test_quaternion_edge_gpu():
    _run("gpu")
