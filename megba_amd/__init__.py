"""megba_amd: MI355X-native distributed bundle adjustment.

A from-scratch reimplementation of the capabilities of MegviiRobot/MegBA
(g2o-style BA problems, Levenberg-Marquardt with distributed
Schur-complement PCG), designed for AMD Instinct MI355X (gfx950):
hand-written HIP kernels for the fused autodiff forward pass, Hessian
assembly, block inverses and the distributed PCG, with RCCL collectives
over xGMI (one process per GPU).
"""
from .problem import BAProblem, load_bal, save_bal  # noqa: F401
from .synth import synthesize_bal  # noqa: F401
from .graph import (  # noqa: F401
    GraphProblem, CameraVertex, PointVertex, ReprojectionEdge)

__version__ = "0.1.0"


def core():
    """Import and return the native core module (built in-tree by build.py)."""
    from . import _core
    return _core
