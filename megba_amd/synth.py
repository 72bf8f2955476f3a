"""Synthetic BAL-shaped problem generator.

No datasets are downloadable in this environment, so benchmarks and tests run
on synthetic problems with the exact shape of the BAL datasets the reference
is quoted on (e.g. Venice problem-1778-993923: 1778 cameras, 993923 points,
~5M observations).  Ground-truth cameras on a ring looking at a point cloud;
observations are noisy projections; the initial estimate is a perturbed
ground truth, which gives LM a realistic multi-iteration trajectory.
"""
import numpy as np


def _look_at_rotation(eye, target, up):
    """World->camera rotation with the camera looking down its -z axis at
    `target` (BAL convention: P_cam = R X + t, visible points have z<0)."""
    fwd = target - eye
    fwd = fwd / np.linalg.norm(fwd)
    # camera -z = fwd  =>  camera z = -fwd
    z = -fwd
    x = np.cross(up, z)
    x = x / np.linalg.norm(x)
    y = np.cross(z, x)
    return np.stack([x, y, z], axis=0)  # rows are camera axes


def _rot_to_angle_axis(R):
    tr = np.clip((np.trace(R) - 1.0) / 2.0, -1.0, 1.0)
    theta = np.arccos(tr)
    if theta < 1e-12:
        return np.zeros(3)
    w = np.array([R[2, 1] - R[1, 2], R[0, 2] - R[2, 0], R[1, 0] - R[0, 1]])
    return theta * w / (2.0 * np.sin(theta))


def synthesize_bal(ncam, npt, nobs, seed=0, pixel_noise=1.0,
                   cam_perturb=2e-3, pt_perturb=1e-2, dtype=np.float64):
    """Returns (cams (ncam,9), pts (npt,3), cam_idx, pt_idx, meas (nobs,2)).

    cams rows: [angle-axis(3), t(3), f, k1, k2].  Guarantees every camera and
    every point has >=2 observations (points need >=2 for a well-conditioned
    Hll block).
    """
    rng = np.random.default_rng(seed)
    radius = 10.0
    # Points: flattened gaussian cloud around origin.
    pts_gt = rng.normal(scale=[3.0, 3.0, 1.5], size=(npt, 3))
    # Cameras on a ring, jittered, looking at the cloud center.
    ang = 2 * np.pi * np.arange(ncam) / max(ncam, 1) + rng.normal(
        scale=0.02, size=ncam)
    eyes = np.stack([radius * np.cos(ang), radius * np.sin(ang),
                     4.0 + rng.normal(scale=0.5, size=ncam)], axis=1)
    cams_gt = np.zeros((ncam, 9))
    Rs = np.zeros((ncam, 3, 3))
    for c in range(ncam):
        R = _look_at_rotation(eyes[c], np.zeros(3), np.array([0.0, 0.0, 1.0]))
        Rs[c] = R
        cams_gt[c, :3] = _rot_to_angle_axis(R)
        cams_gt[c, 3:6] = -R @ eyes[c]
    cams_gt[:, 6] = rng.uniform(800.0, 1200.0, size=ncam)   # f
    cams_gt[:, 7] = rng.normal(scale=1e-7, size=ncam)       # k1
    cams_gt[:, 8] = rng.normal(scale=1e-13, size=ncam)      # k2

    # Observations: ensure degree floors, then fill the rest randomly.
    cam_idx = np.empty(nobs, dtype=np.int32)
    pt_idx = np.empty(nobs, dtype=np.int32)
    base = 0
    # Two observations per point from two distinct cameras.
    need = min(2 * npt, nobs)
    reps = (need + npt - 1) // npt
    pi = np.tile(np.arange(npt, dtype=np.int32), reps)[:need]
    ci = rng.integers(0, ncam, size=need).astype(np.int32)
    # second pass camera differs from first
    if reps >= 2:
        ci[npt:need] = (ci[:need - npt] + 1 +
                        rng.integers(0, max(ncam - 1, 1),
                                     size=need - npt).astype(np.int32)) % ncam
    cam_idx[:need] = ci
    pt_idx[:need] = pi
    base = need
    if base < nobs:
        cam_idx[base:] = rng.integers(0, ncam, size=nobs - base)
        pt_idx[base:] = rng.integers(0, npt, size=nobs - base)
    # Make sure every camera appears at least twice.
    counts = np.bincount(cam_idx, minlength=ncam)
    missing = np.where(counts < 2)[0]
    slot = 0
    for c in missing:
        for _ in range(2 - counts[c]):
            cam_idx[slot] = c
            slot += 1

    # Project with ground truth; observations behind the camera (z >= -0.5,
    # BAL looks down -z) get their CAMERA resampled so point degrees stay
    # intact; a tiny unlucky remainder is depth-clamped.
    def project(idx=None):
        if idx is None:
            return np.einsum('eij,ej->ei', Rs[cam_idx], pts_gt[pt_idx]) \
                + cams_gt[cam_idx, 3:6]
        return np.einsum('eij,ej->ei', Rs[cam_idx[idx]], pts_gt[pt_idx[idx]]) \
            + cams_gt[cam_idx[idx], 3:6]
    P = project()
    for _ in range(8):
        bad = np.where(P[:, 2] > -0.5)[0]
        if len(bad) == 0:
            break
        cam_idx[bad] = rng.integers(0, ncam, size=len(bad))
        P[bad] = project(bad)
    # Re-assert camera degree floor (resampling may have starved a camera).
    counts = np.bincount(cam_idx, minlength=ncam)
    for c in np.where(counts < 2)[0]:
        take = rng.integers(0, nobs, size=2 - counts[c])
        cam_idx[take] = c
        P[take] = project(take)
    P[:, 2] = np.minimum(P[:, 2], -0.5)
    p = -P[:, :2] / P[:, 2:3]
    r2 = (p ** 2).sum(axis=1)
    k1 = cams_gt[cam_idx, 7]
    k2 = cams_gt[cam_idx, 8]
    dist = 1.0 + k1 * r2 + k2 * r2 * r2
    meas = (cams_gt[cam_idx, 6] * dist)[:, None] * p
    meas = meas + rng.normal(scale=pixel_noise, size=meas.shape)

    # Initial estimate: perturbed ground truth.
    cams0 = cams_gt.copy()
    cams0[:, :3] += rng.normal(scale=cam_perturb, size=(ncam, 3))
    cams0[:, 3:6] += rng.normal(scale=10 * cam_perturb, size=(ncam, 3))
    pts0 = pts_gt + rng.normal(scale=pt_perturb, size=(npt, 3))

    return (cams0.astype(dtype), pts0.astype(dtype), cam_idx, pt_idx,
            meas.astype(dtype))
