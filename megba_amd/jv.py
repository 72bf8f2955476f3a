"""Pythonic JetVector wrapper: operator overloading over the native
vectorised dual-number layer, for runtime user-defined edges.

Example (the built-in BAL reprojection edge expressed as a custom forward):

    from megba_amd import jv

    def forward(cam, pt, meas):
        cam, pt, meas = jv.wrap(cam), jv.wrap(pt), jv.wrap(meas)
        R = jv.angle_axis_to_rotation(cam[0:3])
        P = [R[3*i] * pt[0] + R[3*i+1] * pt[1] + R[3*i+2] * pt[2] + cam[3+i]
             for i in range(3)]
        px, py = -P[0] / P[2], -P[1] / P[2]
        fr = jv.radial_distortion([px, py], cam[6:9])
        return ((fr * px - meas[0]).raw, (fr * py - meas[1]).raw)

    problem.build(device="gpu", custom_forward=forward)
"""
from . import _core


class JV:
    __slots__ = ("raw",)

    def __init__(self, raw):
        self.raw = raw

    def _coerce(self, o):
        if isinstance(o, JV):
            return o.raw
        if isinstance(o, (int, float)):
            return _core.jv_scalar(float(o), self.raw.N)
        return o

    def __add__(self, o):
        return JV(_core.jv_add(self.raw, self._coerce(o)))

    def __radd__(self, o):
        return JV(_core.jv_add(self._coerce(o), self.raw))

    def __sub__(self, o):
        return JV(_core.jv_sub(self.raw, self._coerce(o)))

    def __rsub__(self, o):
        return JV(_core.jv_sub(self._coerce(o), self.raw))

    def __mul__(self, o):
        return JV(_core.jv_mul(self.raw, self._coerce(o)))

    def __rmul__(self, o):
        return JV(_core.jv_mul(self._coerce(o), self.raw))

    def __truediv__(self, o):
        return JV(_core.jv_div(self.raw, self._coerce(o)))

    def __rtruediv__(self, o):
        return JV(_core.jv_div(self._coerce(o), self.raw))

    def __neg__(self):
        return JV(_core.jv_neg(self.raw))

    def to_numpy(self):
        return self.raw.to_numpy()


def wrap(seq):
    return [x if isinstance(x, JV) else JV(x) for x in seq]


def sin(a):
    return JV(_core.jv_sin(a.raw))


def cos(a):
    return JV(_core.jv_cos(a.raw))


def sqrt(a):
    return JV(_core.jv_sqrt(a.raw))


def abs(a):  # noqa: A001
    return JV(_core.jv_abs(a.raw))


def normalize_angle(a):
    return JV(_core.jv_normalize_angle(a.raw))


def angle_axis_to_rotation(aa):
    return [JV(r) for r in _core.jv_angle_axis_to_rotation([a.raw for a in aa])]


def rotation2d(theta):
    return [JV(r) for r in _core.jv_rotation2d(theta.raw)]


def quaternion_to_rotation(q):
    return [JV(r) for r in _core.jv_quaternion_to_rotation([a.raw for a in q])]


def normalize_quaternion(q):
    return [JV(r) for r in _core.jv_normalize_quaternion([a.raw for a in q])]


def radial_distortion(p, intr):
    return JV(_core.jv_radial_distortion([a.raw for a in p],
                                         [a.raw for a in intr]))
