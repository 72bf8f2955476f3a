// BAL reprojection residual, generic over the scalar type.
//
// Semantics match the reference user edge (/root/reference/examples/
// BAL_Double.cpp:16-34): camera = [angle-axis(3), t(3), f, k1, k2],
// residual = f * (1 + k1 r^2 + k2 r^4) * p - obs  with  p = -(R x + t) / z.
// This is the standard BAL model (Snavely); the implementation below is the
// textbook Rodrigues form, written fresh for register evaluation (the
// reference evaluates it as a chain of per-op vectorised CUDA kernels).
#pragma once

#include "jet.hpp"

namespace megba {

// Rotate point `pt` by the angle-axis vector `aa` (both arrays of jets or
// scalars); textbook Rodrigues formula with the small-angle guard.
template <typename T, typename JT>
MEGBA_HD inline void angleAxisRotatePoint(const JT aa[3], const JT pt[3],
                                          JT out[3]) {
  const JT theta2 = aa[0] * aa[0] + aa[1] * aa[1] + aa[2] * aa[2];
  if (theta2.v > T(1e-14)) {
    const JT theta = sqrt(theta2);
    const JT costh = cos(theta);
    const JT sinth = sin(theta);
    const JT thetaInv = T(1) / theta;
    const JT w0 = aa[0] * thetaInv;
    const JT w1 = aa[1] * thetaInv;
    const JT w2 = aa[2] * thetaInv;
    const JT wxp0 = w1 * pt[2] - w2 * pt[1];
    const JT wxp1 = w2 * pt[0] - w0 * pt[2];
    const JT wxp2 = w0 * pt[1] - w1 * pt[0];
    const JT wdp = (w0 * pt[0] + w1 * pt[1] + w2 * pt[2]) * (T(1) - costh);
    out[0] = pt[0] * costh + wxp0 * sinth + w0 * wdp;
    out[1] = pt[1] * costh + wxp1 * sinth + w1 * wdp;
    out[2] = pt[2] * costh + wxp2 * sinth + w2 * wdp;
  } else {
    // theta ~ 0: R x ~ x + aa x x
    out[0] = pt[0] + (aa[1] * pt[2] - aa[2] * pt[1]);
    out[1] = pt[1] + (aa[2] * pt[0] - aa[0] * pt[2]);
    out[2] = pt[2] + (aa[0] * pt[1] - aa[1] * pt[0]);
  }
}

// cam[9], pt[3] are jets (leaves seeded by the caller); meas[2] plain scalars.
template <typename T, typename JT>
MEGBA_HD inline void balReprojectionError(const JT cam[9], const JT pt[3],
                                          const T meas[2], JT res[2]) {
  JT P[3];
  angleAxisRotatePoint<T, JT>(cam, pt, P);
  P[0] += cam[3];
  P[1] += cam[4];
  P[2] += cam[5];
  const JT invNegZ = T(-1) / P[2];
  const JT xp = P[0] * invNegZ;
  const JT yp = P[1] * invNegZ;
  const JT r2 = xp * xp + yp * yp;
  const JT distortion = T(1) + r2 * (cam[7] + cam[8] * r2);
  const JT scaled = cam[6] * distortion;
  res[0] = scaled * xp - meas[0];
  res[1] = scaled * yp - meas[1];
}

// (6,3,2) built-in: BAL with per-problem FIXED intrinsics (calibrated
// camera rig) — camera = [angle-axis(3), t(3)], intr = {f, k1, k2} plain
// constants.  Same projection model as BAL, intrinsics out of the state.
template <typename T, typename JT>
MEGBA_HD inline void balFixedIntrError(const JT cam[6], const JT pt[3],
                                       const T meas[2], const T intr[3],
                                       JT res[2]) {
  JT P[3];
  angleAxisRotatePoint<T, JT>(cam, pt, P);
  P[0] += cam[3];
  P[1] += cam[4];
  P[2] += cam[5];
  const JT invNegZ = T(-1) / P[2];
  const JT xp = P[0] * invNegZ;
  const JT yp = P[1] * invNegZ;
  const JT r2 = xp * xp + yp * yp;
  const JT scaled = intr[0] * (T(1) + r2 * (intr[1] + intr[2] * r2));
  res[0] = scaled * xp - meas[0];
  res[1] = scaled * yp - meas[1];
}

// (6,3,3) built-in: SE3 point-alignment residual r = R(aa) p + t - meas
// (3D registration / pose-graph-style landmark edge).
template <typename T, typename JT>
MEGBA_HD inline void se3PointError(const JT cam[6], const JT pt[3],
                                   const T meas[3], JT res[3]) {
  JT P[3];
  angleAxisRotatePoint<T, JT>(cam, pt, P);
  res[0] = P[0] + cam[3] - meas[0];
  res[1] = P[1] + cam[4] - meas[1];
  res[2] = P[2] + cam[5] - meas[2];
}

// Does a built-in residual exist for these dims?  (Other combinations are
// supported via the custom-forward path only.)
MEGBA_HD inline bool hasBuiltinResidual(int cd, int pd, int rd) {
  return (cd == 9 && pd == 3 && rd == 2) || (cd == 6 && pd == 3 && rd == 2) ||
         (cd == 6 && pd == 3 && rd == 3);
}

// Dims-dispatched built-in residual (compile-time selection; `intr` only
// read by the fixed-intrinsics model).
template <typename T, typename JT, int CD, int PD, int RD>
MEGBA_HD inline void builtinResidual(const JT* cam, const JT* pt,
                                     const T* meas, const T* intr, JT* res) {
  if constexpr (CD == 9 && PD == 3 && RD == 2) {
    (void)intr;
    balReprojectionError<T, JT>(cam, pt, meas, res);
  } else if constexpr (CD == 6 && PD == 3 && RD == 2) {
    balFixedIntrError<T, JT>(cam, pt, meas, intr, res);
  } else if constexpr (CD == 6 && PD == 3 && RD == 3) {
    (void)intr;
    se3PointError<T, JT>(cam, pt, meas, res);
  } else {
    // custom-forward-only dims: engines refuse AUTO diff without a custom
    // forward at build time, so this is never reached.
    for (int i = 0; i < RD; ++i) res[i] = JT(T(0));
    (void)cam;
    (void)pt;
    (void)meas;
    (void)intr;
  }
}

}  // namespace megba
