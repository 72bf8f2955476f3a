"""Torch-free TCP rendezvous (megba_amd.rendezvous): the GPU bench path
bootstraps RCCL through this store with no torch.distributed involved.
Exercised here with real processes over localhost."""
import multiprocessing as mp
import os
import struct
import subprocess
import sys

import numpy as np


def _worker(rank, world, port, q):
    try:
        from megba_amd.rendezvous import TcpStore
        st = TcpStore("127.0.0.1", port, rank, world, timeout=30.0)
        # broadcast: rank 0's 128 pseudo-id bytes reach everyone
        payload = bytes(range(128)) if rank == 0 else None
        got = st.broadcast_bytes(payload)
        assert got == bytes(range(128)), "broadcast mismatch"
        st.barrier()
        # second broadcast (the preflight/engine double-exchange pattern)
        payload2 = b"second" if rank == 0 else None
        got2 = st.broadcast_bytes(payload2)
        assert got2 == b"second"
        mx = st.all_max(float(rank) * 1.5)
        assert mx == (world - 1) * 1.5, mx
        st.barrier()
        st.close()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"FAIL: {e!r}"))


def test_store_world4():
    world = 4
    port = 29871
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=60) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    assert all(msg == "ok" for _, msg in results), results


def test_store_world1_noop():
    from megba_amd.rendezvous import TcpStore
    st = TcpStore("127.0.0.1", 29872, 0, 1)
    assert st.broadcast_bytes(b"x") == b"x"
    st.barrier()
    assert st.all_max(3.0) == 3.0
    st.close()


def test_missing_peer_times_out():
    from megba_amd.rendezvous import TcpStore
    import pytest
    with pytest.raises(TimeoutError):
        TcpStore("127.0.0.1", 29873, 0, 2, timeout=1.0)


def test_bench_gpu_path_imports_without_torch():
    """The GPU branch of bench.py must not touch torch: simulate a world-1
    non-distributed run with torch import blocked (device=gpu falls back to
    an error only at engine build on a no-GPU box, which is fine -- the
    bootstrap code itself must get that far without torch)."""
    code = (
        "import sys\n"
        "class _Block:\n"
        "    def find_module(self, name, path=None):\n"
        "        if name == 'torch' or name.startswith('torch.'):\n"
        "            raise ImportError('torch blocked by test')\n"
        "sys.meta_path.insert(0, _Block())\n"
        "import megba_amd\n"
        "from megba_amd.rendezvous import TcpStore\n"
        "st = TcpStore('127.0.0.1', 29874, 0, 1)\n"
        "st.close()\n"
        "print('NO_TORCH_OK')\n"
    )
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120,
                       cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-2000:]
    assert "NO_TORCH_OK" in r.stdout
