"""World-2 coverage for the feature axes not exercised by the other
distributed tests: runtime custom edges and fixed vertices must behave
identically sharded and unsharded."""
import json
import os

import numpy as np

import megba_amd as mb

SEED = 23
SHAPE = (10, 90, 760)


def _forward(cam, pt, meas):
    from megba_amd import jv
    cam, pt, meas = jv.wrap(cam), jv.wrap(pt), jv.wrap(meas)
    R = jv.angle_axis_to_rotation(cam[0:3])
    P = [R[3 * i] * pt[0] + R[3 * i + 1] * pt[1] + R[3 * i + 2] * pt[2]
         + cam[3 + i] for i in range(3)]
    px, py = -P[0] / P[2], -P[1] / P[2]
    fr = jv.radial_distortion([px, py], cam[6:9])
    return ((fr * px - meas[0]).raw, (fr * py - meas[1]).raw)


def _run(rank, world, custom, fixed, allreduce=None):
    cams, pts, ci, pi, meas = mb.synthesize_bal(*SHAPE, seed=SEED)
    cam_fixed = np.zeros(len(cams), dtype=np.uint8)
    cam_fixed[1] = 1
    p = mb.BAProblem(cams, pts, ci, pi, meas,
                     cam_fixed=cam_fixed if fixed else None)
    p.build(device="cpu", rank=rank, world_size=world,
            allreduce=allreduce,
            custom_forward=_forward if custom else None)
    rep = p.solve(max_iter=5, tau=1e4, solver_tol=1e-6, solver_max_iter=150,
                  solver_refuse_ratio=1e6, verbose=False)
    c, q = p.get_params()
    return [it["chi2"] for it in rep["iters"]], c


def _worker(rank, world, port, out, custom, fixed):
    import torch.distributed as dist
    from megba_amd.dist import gloo_allreduce_callback
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{port}",
                            rank=rank, world_size=world)
    try:
        chis, c = _run(rank, world, custom, fixed,
                       gloo_allreduce_callback())
        if rank == 0:
            np.save(out + ".cams.npy", c)
            with open(out, "w") as f:
                json.dump(chis, f)
    finally:
        dist.destroy_process_group()


def _check(tmp_path, custom, fixed, port):
    import torch.multiprocessing as mp
    ref, c1 = _run(0, 1, custom, fixed)
    out = str(tmp_path / f"c{int(custom)}{int(fixed)}.json")
    mp.spawn(_worker, args=(2, port, out, custom, fixed), nprocs=2,
             join=True)
    chis = json.loads(open(out).read())
    np.testing.assert_allclose(chis, ref, rtol=1e-6)
    c2 = np.load(out + ".cams.npy")
    np.testing.assert_allclose(c2, c1, rtol=1e-5, atol=1e-8)
    if fixed:
        cams0 = mb.synthesize_bal(*SHAPE, seed=SEED)[0]
        np.testing.assert_array_equal(c2[1], cams0[1])


def test_world2_custom_forward(tmp_path):
    _check(tmp_path, custom=True, fixed=False, port=29531)


def test_world2_fixed_vertices(tmp_path):
    _check(tmp_path, custom=False, fixed=True, port=29532)
