"""BAL text format round-trip (the reference's problem/checkpoint format,
examples/BAL_Double.cpp:74-139) and the CLI example."""
import subprocess
import sys

import numpy as np

import megba_amd as mb


def test_bal_roundtrip(tmp_path):
    cams, pts, ci, pi, meas = mb.synthesize_bal(8, 50, 400, seed=1)
    f = tmp_path / "prob.txt"
    mb.save_bal(f, cams, pts, ci, pi, meas)
    c2, p2, ci2, pi2, m2 = mb.load_bal(f)
    np.testing.assert_allclose(c2, cams, rtol=1e-15)
    np.testing.assert_allclose(p2, pts, rtol=1e-15)
    np.testing.assert_array_equal(ci2, ci)
    np.testing.assert_array_equal(pi2, pi)
    np.testing.assert_allclose(m2, meas, rtol=1e-15)


def test_cli_runs_cpu(tmp_path):
    cams, pts, ci, pi, meas = mb.synthesize_bal(8, 50, 400, seed=1)
    f = tmp_path / "prob.txt"
    out = tmp_path / "solved.txt"
    mb.save_bal(f, cams, pts, ci, pi, meas)
    r = subprocess.run(
        [sys.executable, "examples/bal_solve.py", "--path", str(f),
         "--device", "cpu", "--max_iter", "4", "--tau", "1e4",
         "--out", str(out)],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert "Iter" in r.stdout and "Finished" in r.stdout
    c2, *_ = mb.load_bal(out)
    assert not np.allclose(c2, cams)


def test_cpp_cli_runs_cpu(tmp_path):
    import os
    binpath = "examples/bal_solve_cpp"
    if not os.path.exists(binpath):
        import pytest
        pytest.skip("native example not built")
    cams, pts, ci, pi, meas = mb.synthesize_bal(8, 50, 400, seed=1)
    f = tmp_path / "prob.txt"
    mb.save_bal(f, cams, pts, ci, pi, meas)
    r = subprocess.run([binpath, "--path", str(f), "--device", "cpu",
                        "--max_iter", "4", "--tau", "1e4"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert "final error" in r.stdout


def test_cli_out_roundtrip(tmp_path):
    """bal_solve.py --out writes a loadable BAL file of the solved state."""
    import subprocess
    import sys
    out = str(tmp_path / "solved.txt")
    r = subprocess.run([sys.executable, "examples/bal_solve.py",
                        "--synthetic", "ladybug", "--device", "cpu",
                        "--max_iter", "2", "--out", out],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-500:]
    cams, pts, ci, pi, meas = mb.load_bal(out)
    assert cams.shape == (49, 9) and pts.shape == (7776, 3)
    assert len(ci) == len(pi) == len(meas)
