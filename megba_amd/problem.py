"""High-level problem API (g2o/MegBA-style) + BAL text IO.

Mirrors the reference's user surface (BaseProblem + BAL_* example flags,
/root/reference/examples/BAL_Double.cpp:50-58): the same knob names
(world_size, max_iter, solver_tol, solver_refuse_ratio, solver_max_iter,
tau, epsilon1, epsilon2) with the same semantics.
"""
import numpy as np


class BAProblem:
    """A BAL-family bundle adjustment problem.

    Block dims are inferred from the array shapes; the compiled set is
    camDim in {9, 6, 4}, ptDim = 3, resDim in {2, 3} (the reference takes
    these as runtime kernel arguments, build_linear_system.cu:48-146).

    cams: (ncam, 9) BAL [angle-axis(3), t(3), f, k1, k2]  -- or (ncam, 6)
          calibrated [angle-axis(3), t(3)] (pass intrinsics= to build()),
          or (ncam, 4) with a custom_forward.
    pts:  (npt, 3)
    cam_idx/pt_idx: (nobs,) int
    meas: (nobs, 2) image observations, or (nobs, 3) for 3D residuals
          (e.g. the built-in (6,3,3) SE3 point-alignment model)
    info: optional (nobs, resDim*(resDim+1)/2) packed-upper symmetric
          information (resDim=2: [w00, w01, w11])
    """

    def __init__(self, cams, pts, cam_idx, pt_idx, meas, info=None,
                 cam_fixed=None, pt_fixed=None):
        from . import _core
        self.cams = np.ascontiguousarray(cams, dtype=np.float64)
        self.pts = np.ascontiguousarray(pts, dtype=np.float64)
        self.cam_idx = np.ascontiguousarray(cam_idx, dtype=np.int32)
        self.pt_idx = np.ascontiguousarray(pt_idx, dtype=np.int32)
        self.meas = np.ascontiguousarray(meas, dtype=np.float64)
        self.info = None if info is None else np.ascontiguousarray(
            info, dtype=np.float64)
        cf = None if cam_fixed is None else np.ascontiguousarray(
            cam_fixed, dtype=np.uint8)
        pf = None if pt_fixed is None else np.ascontiguousarray(
            pt_fixed, dtype=np.uint8)
        self._core = _core.Problem(self.cams, self.pts, self.cam_idx,
                                   self.pt_idx, self.meas, self.info, cf, pf)
        self._built = False

    # -- build -------------------------------------------------------------
    def build(self, device="cpu", dtype="float64", rank=0, world_size=1,
              device_index=0, diff="auto", schur="explicit", loss="none",
              loss_delta=1.0, allreduce=None, rccl_id=None,
              custom_forward=None, intrinsics=None):
        """loss: robust loss ("none" | "huber" | "cauchy") with scale
        loss_delta -- IRLS reweighting, rho-consistent cost (beyond the
        reference, which has only the 2x2 information matrix).
        custom_forward: optional callable (cam_jvs[camDim], pt_jvs[3],
        meas_jvs[resDim]) -> resDim JetVectors, evaluated per forward pass
        (runtime user-defined edges; see megba_amd.jv helpers).
        intrinsics: [f, k1, k2] for the (6,3,2) fixed-intrinsics built-in."""
        self._core.build(device=device, dtype=dtype, rank=rank,
                         world_size=world_size, device_index=device_index,
                         diff=diff, schur=schur, loss=loss,
                         loss_delta=loss_delta, allreduce=allreduce,
                         rccl_id=rccl_id, custom_forward=custom_forward,
                         intrinsics=intrinsics)
        self._built = True
        return self

    # -- solve -------------------------------------------------------------
    def solve(self, max_iter=20, tau=1e4, epsilon1=1.0, epsilon2=1e-10,
              solver_max_iter=100, solver_tol=1e-1, solver_refuse_ratio=1.0,
              force_iterations=False, verbose=True):
        assert self._built, "call build() first"
        return self._core.solve(
            max_iter=max_iter, tau=tau, epsilon1=epsilon1, epsilon2=epsilon2,
            solver_max_iter=solver_max_iter, solver_tol=solver_tol,
            solver_refuse_ratio=solver_refuse_ratio,
            force_iterations=force_iterations, verbose=verbose)

    # -- low-level steps (tests) -------------------------------------------
    def __getattr__(self, name):
        # Delegate fine-grained methods to the core object.
        if name.startswith("_"):
            raise AttributeError(name)
        return getattr(self._core, name)

    def params(self):
        """Solved parameters (cams, pts).  COLLECTIVE when world_size>1:
        every rank must call it (the point shards are merged with an
        allreduce)."""
        return self._core.get_params()


def load_bal(path):
    """Parse a BAL 'problem-*.txt' file (same format the reference examples
    read, /root/reference/examples/BAL_Double.cpp:74-139)."""
    with open(path) as f:
        header = f.readline().split()
        ncam, npt, nobs = int(header[0]), int(header[1]), int(header[2])
        text = f.read()
    try:
        body = np.fromstring(text, sep=" ")
    except Exception:  # numpy versions without text-mode fromstring
        body = np.array(text.split(), dtype=np.float64)
    obs = body[:nobs * 4].reshape(nobs, 4)
    cam_idx = obs[:, 0].astype(np.int32)
    pt_idx = obs[:, 1].astype(np.int32)
    meas = obs[:, 2:4].copy()
    rest = body[nobs * 4:]
    cams = rest[:ncam * 9].reshape(ncam, 9).copy()
    pts = rest[ncam * 9: ncam * 9 + npt * 3].reshape(npt, 3).copy()
    return cams, pts, cam_idx, pt_idx, meas


def save_bal(path, cams, pts, cam_idx, pt_idx, meas):
    ncam, npt, nobs = len(cams), len(pts), len(cam_idx)
    with open(path, "w") as f:
        f.write(f"{ncam} {npt} {nobs}\n")
        for i in range(nobs):
            f.write(f"{cam_idx[i]} {pt_idx[i]} {meas[i,0]:.17g} {meas[i,1]:.17g}\n")
        for row in np.asarray(cams).reshape(-1):
            f.write(f"{row:.17g}\n")
        for row in np.asarray(pts).reshape(-1):
            f.write(f"{row:.17g}\n")
