"""g2o-style incremental graph-construction API.

The reference's user-facing surface is a vertex/edge graph built one
element at a time (BaseProblem::appendVertex / appendEdge,
/root/reference/include/problem/base_problem.h:22-83; vertex/edge classes
/root/reference/include/vertex/base_vertex.h:27-230 and
/root/reference/include/edge/base_edge.h:26-163; usage pattern
/root/reference/examples/BAL_Double.cpp:60-164).  This module provides the
same construction style on top of the array-based `BAProblem` core: append
vertices and edges, `solve()`, then read each vertex's `.estimation`
(the reference's writeBack semantics, base_problem.cpp:250-272).
"""
import numpy as np

from .problem import BAProblem

CAMERA = 0
POINT = 1


class BaseVertex:
    """A parameter block.  kind: CAMERA (9|6|4 params — the compiled
    camDim set; see BAProblem) or POINT (3)."""

    def __init__(self, estimation, kind, fixed=False):
        est = np.asarray(estimation, dtype=np.float64).reshape(-1)
        ok = est.size in (9, 6, 4) if kind == CAMERA else est.size == 3
        if not ok:
            raise ValueError(f"vertex kind {kind} needs "
                             f"{'9|6|4' if kind == CAMERA else '3'} params, "
                             f"got {est.size}")
        self.estimation = est.copy()
        self.kind = kind
        self.fixed = bool(fixed)
        self._slot = None  # assigned by GraphProblem.append_vertex


class CameraVertex(BaseVertex):
    def __init__(self, estimation, fixed=False):
        super().__init__(estimation, CAMERA, fixed)


class PointVertex(BaseVertex):
    def __init__(self, estimation, fixed=False):
        super().__init__(estimation, POINT, fixed)


class ReprojectionEdge:
    """One observation: connect exactly one CameraVertex and one
    PointVertex (the reference only implements the 1-camera-1-point edge
    kind, base_edge.cpp:27-36), with a 2-d measurement and an optional
    symmetric 2x2 information matrix given as (i00, i01, i11)."""

    def __init__(self, measurement, information=None):
        m = np.asarray(measurement, dtype=np.float64).reshape(-1)
        if m.size != 2:
            raise ValueError("measurement must be 2-d")
        self.measurement = m.copy()
        if information is not None:
            information = np.asarray(information, dtype=np.float64).reshape(-1)
            if information.size != 3:
                raise ValueError("information must be (i00, i01, i11)")
        self.information = information
        self.vertices = []

    def append_vertex(self, v):
        if not isinstance(v, BaseVertex):
            raise TypeError("append_vertex expects a BaseVertex")
        self.vertices.append(v)
        return self


class GraphProblem:
    """g2o-style problem graph.  append_vertex / append_edge / solve;
    after solve() every vertex's .estimation holds the optimized value."""

    def __init__(self):
        self._cams = []
        self._pts = []
        self._edges = []

    def append_vertex(self, v):
        if not isinstance(v, BaseVertex):
            raise TypeError("append_vertex expects a BaseVertex")
        if v._slot is not None:
            raise ValueError("vertex already appended")
        if v.kind == CAMERA:
            v._slot = len(self._cams)
            self._cams.append(v)
        else:
            v._slot = len(self._pts)
            self._pts.append(v)
        return self

    def append_edge(self, e):
        kinds = sorted(v.kind for v in e.vertices)
        if kinds != [CAMERA, POINT]:
            raise ValueError("edge must connect exactly one CameraVertex "
                             "and one PointVertex")
        for v in e.vertices:
            if v._slot is None:
                self.append_vertex(v)
        self._edges.append(e)
        return self

    @property
    def n_vertices(self):
        return len(self._cams) + len(self._pts)

    @property
    def n_edges(self):
        return len(self._edges)

    def _assemble(self):
        if not self._edges:
            raise ValueError("no edges")
        cams = np.stack([v.estimation for v in self._cams])
        pts = np.stack([v.estimation for v in self._pts])
        nobs = len(self._edges)
        ci = np.empty(nobs, dtype=np.int32)
        pi = np.empty(nobs, dtype=np.int32)
        meas = np.empty((nobs, 2))
        any_info = any(e.information is not None for e in self._edges)
        info = np.zeros((nobs, 3)) if any_info else None
        if any_info:
            info[:, 0] = 1.0
            info[:, 2] = 1.0  # identity default for unweighted edges
        for k, e in enumerate(self._edges):
            for v in e.vertices:
                if v.kind == CAMERA:
                    ci[k] = v._slot
                else:
                    pi[k] = v._slot
            meas[k] = e.measurement
            if any_info and e.information is not None:
                info[k] = e.information
        cam_fixed = np.array([v.fixed for v in self._cams], dtype=np.uint8)
        pt_fixed = np.array([v.fixed for v in self._pts], dtype=np.uint8)
        kw = {}
        if cam_fixed.any():
            kw["cam_fixed"] = cam_fixed
        if pt_fixed.any():
            kw["pt_fixed"] = pt_fixed
        return BAProblem(cams, pts, ci, pi, meas, info=info, **kw)

    def solve(self, device="cpu", dtype="float64", diff="auto",
              schur="explicit", loss="none", loss_delta=1.0,
              custom_forward=None, intrinsics=None, **solve_kw):
        """Build, run LM, and write the result back into the vertices.
        solve_kw: max_iter, tau, epsilon1, epsilon2, solver_tol,
        solver_max_iter, solver_refuse_ratio, verbose (see BAProblem.solve).
        custom_forward/intrinsics: as in BAProblem.build (user-defined
        residuals; fixed intrinsics for 6-dof cameras).
        """
        p = self._assemble()
        p.build(device=device, dtype=dtype, diff=diff, schur=schur,
                loss=loss, loss_delta=loss_delta,
                custom_forward=custom_forward, intrinsics=intrinsics)
        report = p.solve(**solve_kw)
        cams, pts = p.get_params()
        for v, row in zip(self._cams, cams):
            v.estimation[:] = row
        for v, row in zip(self._pts, pts):
            v.estimation[:] = row
        return report
