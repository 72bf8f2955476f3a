"""Generic block dimensions (round-2 generalization; the reference takes
cameraDim/pointDim/resDim as runtime kernel arguments,
/root/reference/src/edge/build_linear_system.cu:48-146 and common.h:27-46).

Covers the compiled set beyond BAL (9,3,2):
 * (6,3,2) built-in: calibrated BAL (fixed intrinsics) -- Jacobians vs
   finite differences, LM chi2 decrease, and equivalence with the (9,3,2)
   model evaluated at the same parameters.
 * (6,3,3) built-in: SE3 point alignment -- FD check + solve to near-zero.
 * (4,3,2) custom forward: a weak-perspective model expressed in JetVector
   ops -- FD check + LM decrease (no built-in exists for these dims).
 * guards: unsupported dims / missing custom forward raise.
"""
import numpy as np
import pytest

import megba_amd as mb
from megba_amd import jv

INTR = [420.0, -1e-7, 3e-13]


def _synth_calibrated(ncam=10, npt=90, nobs=800, seed=4):
    """Calibrated-camera problem: cams (ncam,6), shared intrinsics INTR."""
    RNG = np.random.default_rng(seed + 100)
    cams9, pts, ci, pi, meas = mb.synthesize_bal(ncam, npt, nobs, seed=seed)
    cams9[:, 6] = INTR[0]
    cams9[:, 7] = INTR[1]
    cams9[:, 8] = INTR[2]
    # recompute measurements for the forced intrinsics via the CPU engine
    p = mb.BAProblem(cams9, pts, ci, pi, np.zeros_like(meas))
    p.build(device="cpu")
    p.forward()
    # dump() returns r in the (pt,cam)-sorted edge order; map back to the
    # original observation order through perm.
    r_sorted = p.dump()["r"].reshape(-1, 2)
    perm = p.index_info()["perm"]
    r = np.empty_like(r_sorted)
    r[perm] = r_sorted
    noise = RNG.normal(0, 0.5, r.shape)
    return cams9, pts, ci, pi, r + noise


def _fd_jacobian(fun, x, eps=1e-6):
    f0 = fun(x)
    J = np.zeros((f0.size, x.size))
    for i in range(x.size):
        xp = x.copy()
        xp[i] += eps
        xm = x.copy()
        xm[i] -= eps
        J[:, i] = (fun(xp) - fun(xm)) / (2 * eps)
    return J


def _bal_fixed_intr_res(cam6, pt, meas2):
    aa, t = cam6[:3], cam6[3:6]
    th = np.linalg.norm(aa)
    if th > 1e-12:
        w = aa / th
        P = (pt * np.cos(th) + np.cross(w, pt) * np.sin(th)
             + w * np.dot(w, pt) * (1 - np.cos(th)))
    else:
        P = pt + np.cross(aa, pt)
    P = P + t
    p2 = -P[:2] / P[2]
    r2 = p2 @ p2
    f, k1, k2 = INTR
    return f * (1 + k1 * r2 + k2 * r2 * r2) * p2 - meas2


def test_632_jacobian_vs_fd():
    RNG = np.random.default_rng(1)
    cams9, pts, ci, pi, meas = _synth_calibrated()
    cams6 = cams9[:, :6].copy()
    p = mb.BAProblem(cams6, pts, ci, pi, meas)
    p.build(device="cpu", intrinsics=INTR)
    p.forward()
    d = p.dump()
    ii = p.index_info()
    nL = len(ci)
    r = d["r"].reshape(nL, 2)
    Jc = d["Jc"].reshape(nL, 2, 6)
    Jp = d["Jp"].reshape(nL, 2, 3)
    for e in RNG.choice(nL, 12, replace=False):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        m = p.meas[ii["perm"][e]]
        np.testing.assert_allclose(
            r[e], _bal_fixed_intr_res(cams6[c], pts[q], m), rtol=1e-9,
            atol=1e-12)
        Jfd_c = _fd_jacobian(
            lambda x: _bal_fixed_intr_res(x, pts[q], m), cams6[c].copy())
        Jfd_p = _fd_jacobian(
            lambda x: _bal_fixed_intr_res(cams6[c], x, m), pts[q].copy())
        np.testing.assert_allclose(Jc[e], Jfd_c, rtol=1e-5, atol=1e-6)
        np.testing.assert_allclose(Jp[e], Jfd_p, rtol=1e-5, atol=1e-6)


def test_632_matches_932_with_same_params():
    """The (6,3,2) model at intrinsics INTR must produce the same residuals
    and the camera-pose J columns of the (9,3,2) model."""
    cams9, pts, ci, pi, meas = _synth_calibrated()
    p9 = mb.BAProblem(cams9, pts, ci, pi, meas)
    p9.build(device="cpu")
    chi9 = p9.forward()
    p6 = mb.BAProblem(cams9[:, :6].copy(), pts, ci, pi, meas)
    p6.build(device="cpu", intrinsics=INTR)
    chi6 = p6.forward()
    np.testing.assert_allclose(chi6, chi9, rtol=1e-12)
    d9, d6 = p9.dump(), p6.dump()
    nL = len(ci)
    np.testing.assert_allclose(d6["r"], d9["r"], rtol=1e-12)
    np.testing.assert_allclose(d6["Jc"].reshape(nL, 2, 6),
                               d9["Jc"].reshape(nL, 2, 9)[:, :, :6],
                               rtol=1e-12)
    np.testing.assert_allclose(d6["Jp"], d9["Jp"], rtol=1e-12)


def test_632_lm_decreases():
    RNG = np.random.default_rng(2)
    cams9, pts, ci, pi, meas = _synth_calibrated(14, 130, 1200, seed=8)
    cams6 = cams9[:, :6] + RNG.normal(0, 0.01, (14, 6))  # start off-optimum
    p = mb.BAProblem(cams6, pts, ci, pi, meas)
    p.build(device="cpu", intrinsics=INTR)
    rep = p.solve(max_iter=8, solver_tol=1e-8, solver_max_iter=200,
                  solver_refuse_ratio=1e6, verbose=False)
    chis = [i["chi2"] for i in rep["iters"]]
    assert chis[-1] < 0.5 * chis[0]
    c2, p2 = p.get_params()
    assert c2.shape == (14, 6) and p2.shape == (130, 3)


def _synth_se3(ncam=8, npt=70, nobs=500, seed=6):
    RNG = np.random.default_rng(seed + 200)
    cams = np.zeros((ncam, 6))
    cams[:, :3] = RNG.normal(0, 0.3, (ncam, 3))
    cams[:, 3:] = RNG.normal(0, 1.0, (ncam, 3))
    pts = RNG.normal(0, 2.0, (npt, 3))
    ci = RNG.integers(0, ncam, nobs).astype(np.int32)
    pi = np.concatenate(
        [np.arange(npt), RNG.integers(0, npt, nobs - npt)]).astype(np.int32)
    ci[:ncam] = np.arange(ncam)

    def se3(cam6, pt):
        aa, t = cam6[:3], cam6[3:]
        th = np.linalg.norm(aa)
        if th > 1e-12:
            w = aa / th
            P = (pt * np.cos(th) + np.cross(w, pt) * np.sin(th)
                 + w * np.dot(w, pt) * (1 - np.cos(th)))
        else:
            P = pt + np.cross(aa, pt)
        return P + t

    meas = np.array([se3(cams[c], pts[q]) for c, q in zip(ci, pi)])
    meas += RNG.normal(0, 0.01, meas.shape)
    return cams, pts, ci, pi, meas, se3


def test_633_se3_fd_and_solve():
    RNG = np.random.default_rng(3)
    cams, pts, ci, pi, meas, se3 = _synth_se3()
    # perturb initial state
    cams0 = cams + RNG.normal(0, 0.02, cams.shape)
    p = mb.BAProblem(cams0, pts, ci, pi, meas)
    p.build(device="cpu")
    chi0 = p.forward()
    d = p.dump()
    ii = p.index_info()
    nL = len(ci)
    Jc = d["Jc"].reshape(nL, 3, 6)
    Jp = d["Jp"].reshape(nL, 3, 3)
    for e in RNG.choice(nL, 8, replace=False):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        m = meas[ii["perm"][e]]
        Jfd_c = _fd_jacobian(lambda x: se3(x, pts[q]) - m, cams0[c].copy())
        Jfd_p = _fd_jacobian(lambda x: se3(cams0[c], x) - m, pts[q].copy())
        np.testing.assert_allclose(Jc[e], Jfd_c, rtol=1e-5, atol=1e-7)
        np.testing.assert_allclose(Jp[e], Jfd_p, rtol=1e-5, atol=1e-7)
    rep = p.solve(max_iter=8, solver_tol=1e-10, solver_max_iter=200,
                  solver_refuse_ratio=1e6, verbose=False)
    assert rep["final_chi2"] < 0.1 * chi0


# ---- (4,3,2): custom forward (weak-perspective-ish model) -----------------
def _wp_forward(cam, pt, meas):
    """cam = [tx, ty, tz, log_f]; r = exp(log_f) * (pt_xy + t_xy) / (pt_z
    + tz + 5) - meas (a contrived but smooth 4-dof camera)."""
    cam, pt, meas = jv.wrap(cam), jv.wrap(pt), jv.wrap(meas)
    # exp(x) via sqrt-free ops: use 1 + x + x^2/2 + x^3/6 (x stays small)
    x = cam[3]
    f = 1.0 + x + x * x * 0.5 + x * x * x * (1.0 / 6.0)
    z = pt[2] + cam[2] + 5.0
    return ((f * (pt[0] + cam[0]) / z - meas[0]).raw,
            (f * (pt[1] + cam[1]) / z - meas[1]).raw)


def _wp_np(cam4, pt, meas2):
    x = cam4[3]
    f = 1.0 + x + x * x / 2 + x ** 3 / 6
    z = pt[2] + cam4[2] + 5.0
    return np.array([f * (pt[0] + cam4[0]) / z - meas2[0],
                     f * (pt[1] + cam4[1]) / z - meas2[1]])


def test_432_custom_forward_fd_and_solve():
    RNG = np.random.default_rng(4)
    ncam, npt, nobs = 6, 50, 360
    cams = RNG.normal(0, 0.1, (ncam, 4))
    pts = RNG.normal(0, 1.0, (npt, 3))
    ci = RNG.integers(0, ncam, nobs).astype(np.int32)
    pi = np.concatenate(
        [np.arange(npt), RNG.integers(0, npt, nobs - npt)]).astype(np.int32)
    ci[:ncam] = np.arange(ncam)
    meas = np.array([_wp_np(cams[c], pts[q], np.zeros(2))
                     for c, q in zip(ci, pi)])
    meas += RNG.normal(0, 0.01, meas.shape)
    # start away from the optimum so LM has something to do
    cams = cams + RNG.normal(0, 0.05, cams.shape)
    pts = pts + RNG.normal(0, 0.05, pts.shape)
    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu", custom_forward=_wp_forward)
    chi0 = p.forward()
    d = p.dump()
    ii = p.index_info()
    nL = nobs
    Jc = d["Jc"].reshape(nL, 2, 4)
    for e in RNG.choice(nL, 8, replace=False):
        c, q = ii["cam_of"][e], ii["pt_of"][e]
        m = meas[ii["perm"][e]]
        Jfd = _fd_jacobian(lambda x: _wp_np(x, pts[q], m), cams[c].copy())
        np.testing.assert_allclose(Jc[e], Jfd, rtol=1e-5, atol=1e-7)
    rep = p.solve(max_iter=6, solver_tol=1e-10, solver_max_iter=150,
                  solver_refuse_ratio=1e6, verbose=False)
    assert rep["final_chi2"] < 0.2 * chi0


def test_unsupported_dims_raise():
    cams = np.zeros((4, 4))
    pts = np.zeros((5, 3))
    ci = np.zeros(6, dtype=np.int32)
    pi = np.arange(5, dtype=np.int32).tolist() + [0]
    meas = np.zeros((6, 2))
    p = mb.BAProblem(cams, pts, ci, np.asarray(pi, np.int32), meas)
    # (4,3,2) has no built-in: must demand a custom forward
    with pytest.raises(RuntimeError, match="custom_forward"):
        p.build(device="cpu")
    # camDim outside the compiled set is refused at construction
    with pytest.raises(RuntimeError, match="9|6|4"):
        mb.BAProblem(np.zeros((4, 7)), pts, ci, np.asarray(pi, np.int32),
                     meas)


def test_analytical_requires_bal_dims():
    cams9, pts, ci, pi, meas = _synth_calibrated()
    p = mb.BAProblem(cams9[:, :6].copy(), pts, ci, pi, meas)
    with pytest.raises(RuntimeError, match="analytical"):
        p.build(device="cpu", diff="analytical", intrinsics=INTR)


# ---- GPU vs CPU-oracle parity for the generic dims ------------------------
def _gpu_cpu_pair(cams, pts, ci, pi, meas, schur="explicit", **kw):
    pc = mb.BAProblem(cams, pts, ci, pi, meas)
    pc.build(device="cpu", schur=schur, **kw)
    pg = mb.BAProblem(cams, pts, ci, pi, meas)
    pg.build(device="gpu", schur=schur, **kw)
    return pc, pg


def _compare_stage(pc, pg, keys=("r", "Jc", "Jp", "Hpp", "Hll", "g"),
                   rtol=1e-9):
    c1, c2 = pc.forward(), pg.forward()
    np.testing.assert_allclose(c2, c1, rtol=rtol)
    dj1, dj2 = pc.dump(), pg.dump()
    pc.accept_forward()
    pg.accept_forward()
    pc.build_linear_system()
    pg.build_linear_system()
    d1, d2 = pc.dump(), pg.dump()
    d1.update({k: dj1[k] for k in ("r", "Jc", "Jp")})
    d2.update({k: dj2[k] for k in ("r", "Jc", "Jp")})
    for key in keys:
        scale = np.abs(d1[key]).max() or 1.0
        np.testing.assert_allclose(d2[key], d1[key], rtol=rtol,
                                   atol=rtol * scale, err_msg=key)
    # one damped solve must agree too
    pc.process_diag(1e4)
    pg.process_diag(1e4)
    pc.solve_linear(max_iter=40, tol=1e-12, refuse_ratio=1e30)
    pg.solve_linear(max_iter=40, tol=1e-12, refuse_ratio=1e30)
    dx1 = pc.dump()["deltaX"]
    dx2 = pg.dump()["deltaX"]
    scale = np.abs(dx1).max() or 1.0
    np.testing.assert_allclose(dx2, dx1, rtol=1e-6, atol=1e-8 * scale,
                               err_msg="deltaX")


@pytest.mark.gpu
@pytest.mark.parametrize("schur", ["explicit", "implicit"])
def test_gpu_632_matches_cpu_oracle(schur):
    cams9, pts, ci, pi, meas = _synth_calibrated(12, 110, 950, seed=12)
    pc, pg = _gpu_cpu_pair(cams9[:, :6].copy(), pts, ci, pi, meas,
                           schur=schur, intrinsics=INTR)
    _compare_stage(pc, pg)


@pytest.mark.gpu
@pytest.mark.parametrize("schur", ["explicit", "implicit"])
def test_gpu_633_matches_cpu_oracle(schur):
    cams, pts, ci, pi, meas, _ = _synth_se3(9, 80, 600, seed=21)
    pc, pg = _gpu_cpu_pair(cams, pts, ci, pi, meas, schur=schur)
    _compare_stage(pc, pg)


@pytest.mark.gpu
def test_gpu_432_custom_matches_cpu_oracle():
    RNG = np.random.default_rng(31)
    ncam, npt, nobs = 6, 50, 360
    cams = RNG.normal(0, 0.1, (ncam, 4))
    pts = RNG.normal(0, 1.0, (npt, 3))
    ci = RNG.integers(0, ncam, nobs).astype(np.int32)
    pi = np.concatenate(
        [np.arange(npt), RNG.integers(0, npt, nobs - npt)]).astype(np.int32)
    ci[:ncam] = np.arange(ncam)
    meas = np.array([_wp_np(cams[c], pts[q], np.zeros(2))
                     for c, q in zip(ci, pi)])
    pc, pg = _gpu_cpu_pair(cams, pts, ci, pi, meas,
                           custom_forward=_wp_forward)
    _compare_stage(pc, pg)


@pytest.mark.gpu
def test_gpu_632_lm_trajectory_matches_cpu():
    RNG = np.random.default_rng(41)
    cams9, pts, ci, pi, meas = _synth_calibrated(12, 110, 950, seed=13)
    cams6 = cams9[:, :6] + RNG.normal(0, 0.01, (12, 6))
    kw = dict(max_iter=5, solver_tol=1e-8, solver_max_iter=150,
              solver_refuse_ratio=1e6, verbose=False)
    pc = mb.BAProblem(cams6, pts, ci, pi, meas)
    pc.build(device="cpu", intrinsics=INTR)
    pg = mb.BAProblem(cams6, pts, ci, pi, meas)
    pg.build(device="gpu", intrinsics=INTR)
    r1, r2 = pc.solve(**kw), pg.solve(**kw)
    np.testing.assert_allclose([i["chi2"] for i in r2["iters"]],
                               [i["chi2"] for i in r1["iters"]], rtol=1e-5)


@pytest.mark.gpu
def test_gpu_633_fp32_decreases():
    cams, pts, ci, pi, meas, _ = _synth_se3(9, 80, 600, seed=22)
    RNG = np.random.default_rng(5)
    cams0 = cams + RNG.normal(0, 0.02, cams.shape)
    p = mb.BAProblem(cams0, pts, ci, pi, meas)
    p.build(device="gpu", dtype="float32")
    rep = p.solve(max_iter=6, solver_tol=1e-6, solver_max_iter=100,
                  solver_refuse_ratio=1e6, verbose=False)
    chis = [i["chi2"] for i in rep["iters"]]
    assert chis[-1] < 0.5 * chis[0]


def test_graph_api_6dof_camera():
    """g2o-style graph construction with 6-dof (calibrated) cameras."""
    from megba_amd.graph import GraphProblem, CameraVertex, PointVertex, \
        ReprojectionEdge
    cams9, pts, ci, pi, meas = _synth_calibrated(8, 60, 500, seed=19)
    g = GraphProblem()
    cvs = [CameraVertex(cams9[i, :6]) for i in range(8)]
    pvs = [PointVertex(pts[i]) for i in range(60)]
    for c, q, m in zip(ci, pi, meas):
        e = ReprojectionEdge(m)
        e.append_vertex(cvs[c]).append_vertex(pvs[q])
        g.append_edge(e)
    rep = g.solve(device="cpu", intrinsics=INTR, max_iter=4,
                  solver_tol=1e-8, solver_max_iter=100,
                  solver_refuse_ratio=1e6, verbose=False)
    chis = [i["chi2"] for i in rep["iters"]]
    assert chis[-1] <= chis[0]
    assert cvs[0].estimation.shape == (6,)


def _dims_worker(rank, world_size, port, out_path):
    import json
    import torch.distributed as dist
    from megba_amd.dist import gloo_allreduce_callback
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world_size)
    try:
        cams9, pts, ci, pi, meas = _synth_calibrated(14, 130, 1200, seed=8)
        p = mb.BAProblem(cams9[:, :6].copy(), pts, ci, pi, meas)
        p.build(device="cpu", rank=rank, world_size=world_size,
                allreduce=gloo_allreduce_callback(), intrinsics=INTR)
        rep = p.solve(max_iter=6, tau=1e4, solver_tol=1e-8,
                      solver_max_iter=150, solver_refuse_ratio=1e6,
                      verbose=False)
        c2, p2 = p.get_params()  # collective
        if rank == 0:
            with open(out_path, "w") as f:
                json.dump([i["chi2"] for i in rep["iters"]], f)
    finally:
        dist.destroy_process_group()


def test_632_world2_matches_world1(tmp_path):
    """Generic dims under distribution: the point-sharded (6,3,2) engine
    at world_size=2 (gloo) must reproduce the single-process trajectory —
    same reduction points as the RCCL GPU path."""
    import json
    import torch.multiprocessing as mp
    cams9, pts, ci, pi, meas = _synth_calibrated(14, 130, 1200, seed=8)
    p1 = mb.BAProblem(cams9[:, :6].copy(), pts, ci, pi, meas)
    p1.build(device="cpu", intrinsics=INTR)
    rep1 = p1.solve(max_iter=6, tau=1e4, solver_tol=1e-8,
                    solver_max_iter=150, solver_refuse_ratio=1e6,
                    verbose=False)
    out = tmp_path / "w2.json"
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_dims_worker, args=(r, 2, 29733, str(out)))
             for r in range(2)]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=300)
        assert pr.exitcode == 0
    chis2 = json.loads(out.read_text())
    chis1 = [i["chi2"] for i in rep1["iters"]]
    np.testing.assert_allclose(chis2, chis1, rtol=1e-8)
