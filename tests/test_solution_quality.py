"""Cross-solver quality check: our LM must reach the same cost as scipy's
trust-region least_squares on the same problem (independent implementation,
independent Jacobians)."""
import numpy as np
import pytest

import megba_amd as mb
from conftest import bal_residual_np


def test_matches_scipy_least_squares():
    from scipy.optimize import least_squares
    from scipy.sparse import lil_matrix

    cams, pts, ci, pi, meas = mb.synthesize_bal(6, 40, 320, seed=5)
    ncam, npt, nobs = len(cams), len(pts), len(ci)

    p = mb.BAProblem(cams, pts, ci, pi, meas)
    p.build(device="cpu")
    rep = p.solve(max_iter=30, tau=1e4, epsilon1=1e-10, epsilon2=1e-14,
                  solver_tol=1e-10, solver_max_iter=500,
                  solver_refuse_ratio=1e9, verbose=False)
    ours = rep["final_chi2"]

    def residuals(x):
        c = x[:ncam * 9].reshape(ncam, 9)
        q = x[ncam * 9:].reshape(npt, 3)
        out = np.empty(2 * nobs)
        for e in range(nobs):
            out[2 * e:2 * e + 2] = bal_residual_np(c[ci[e]], q[pi[e]], meas[e])
        return out

    spar = lil_matrix((2 * nobs, 9 * ncam + 3 * npt), dtype=int)
    for e in range(nobs):
        spar[2 * e:2 * e + 2, 9 * ci[e]:9 * ci[e] + 9] = 1
        spar[2 * e:2 * e + 2, 9 * ncam + 3 * pi[e]:9 * ncam + 3 * pi[e] + 3] = 1

    x0 = np.concatenate([cams.reshape(-1), pts.reshape(-1)])
    res = least_squares(residuals, x0, jac_sparsity=spar, method="trf",
                        max_nfev=60, xtol=1e-12, ftol=1e-12)
    scipy_chi2 = 2 * res.cost  # scipy cost = 0.5 * sum r^2

    # Both should land at (essentially) the same local optimum.
    assert ours <= scipy_chi2 * 1.02 + 1e-9, (ours, scipy_chi2)
