"""Two ranks sharing ONE GPU via the gloo host-allreduce fallback: exercises
the full point-sharded GPU code path (partial assembly, local point side,
camera-vector collectives) with world_size=2 on a single device — in both
Schur modes (implicit is the flagship bench default)."""
import json

import numpy as np
import pytest

SEED = 11
SHAPE = (15, 160, 1400)

pytestmark = pytest.mark.gpu


def _worker(rank, world_size, port, out_path, schur):
    import torch.distributed as dist
    from megba_amd.dist import gloo_allreduce_callback
    dist.init_process_group("gloo", init_method=f"tcp://127.0.0.1:{port}",
                            rank=rank, world_size=world_size)
    try:
        import megba_amd as mb
        cams, pts, ci, pi, meas = mb.synthesize_bal(*SHAPE, seed=SEED)
        p = mb.BAProblem(cams, pts, ci, pi, meas)
        p.build(device="gpu", rank=rank, world_size=world_size,
                device_index=0, schur=schur,
                allreduce=gloo_allreduce_callback())
        rep = p.solve(max_iter=6, tau=1e4, solver_tol=1e-6,
                      solver_max_iter=200, solver_refuse_ratio=1e6,
                      verbose=False)
        chis = [it["chi2"] for it in rep["iters"]]
        c2, p2 = p.get_params()  # collective
        if rank == 0:
            np.save(out_path + ".pts.npy", p2)
            with open(out_path, "w") as f:
                json.dump(chis, f)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("schur,port", [("explicit", 29521),
                                        ("implicit", 29522)])
def test_gpu_world2_single_device(tmp_path, schur, port):
    import megba_amd as mb
    import torch.multiprocessing as mp
    cams, pts, ci, pi, meas = mb.synthesize_bal(*SHAPE, seed=SEED)
    p1 = mb.BAProblem(cams, pts, ci, pi, meas)
    p1.build(device="gpu", schur=schur)
    rep = p1.solve(max_iter=6, tau=1e4, solver_tol=1e-6, solver_max_iter=200,
                   solver_refuse_ratio=1e6, verbose=False)
    ref = [it["chi2"] for it in rep["iters"]]
    _, q1 = p1.get_params()
    out = tmp_path / f"chis_{schur}.json"
    mp.spawn(_worker, args=(2, port, str(out), schur), nprocs=2, join=True)
    chis = json.loads(out.read_text())
    assert len(chis) == len(ref)
    np.testing.assert_allclose(chis, ref, rtol=1e-6)
    q2 = np.load(str(out) + ".pts.npy")
    np.testing.assert_allclose(q2, q1, rtol=1e-6, atol=1e-9)
