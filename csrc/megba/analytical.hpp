// Closed-form BAL reprojection residual + 2x12 Jacobian (no dual numbers).
//
// Capability anchor: the reference's analytical-derivatives module
// (/root/reference/src/geo/analytical_derivatives.cu:16-322) which computes
// the fused residual + Jacobian in one kernel and is its performance
// flagship (-30% time vs autodiff, README.md:16).  Derivation here is the
// standard Rodrigues differential:
//   P = c X + s (w x X) + (1-c)(w.X) w + t,  w = aa/theta
//   dP/daa = A w^T + [ s (-[X]x) + (1-c)(w X^T + (w.X) I) ] (I - w w^T)/theta
//   with A = -s X + c (w x X) + s (w.X) w
// then chained through the perspective divide and radial distortion.
// Validated against the autodiff (Jet) path in tests/test_analytical.py.
#pragma once

#include "jet.hpp"  // MEGBA_HD

namespace megba {

template <typename T>
MEGBA_HD inline void balAnalytical(const T cam[9], const T pt[3],
                                   const T meas[2], T res[2], T Jc[2][9],
                                   T Jp[2][3]) {
  const T ax = cam[0], ay = cam[1], az = cam[2];
  const T X0 = pt[0], X1 = pt[1], X2 = pt[2];
  const T theta2 = ax * ax + ay * ay + az * az;

  T P[3];        // rotated point (before +t)
  T R[3][3];     // dP/dX
  T dPda[3][3];  // dP/daa
  if (theta2 > T(1e-14)) {
#ifdef __HIP_DEVICE_COMPILE__
    const T theta = ::sqrt(theta2);
    const T c = ::cos(theta);
    const T s = ::sin(theta);
#else
    const T theta = std::sqrt(theta2);
    const T c = std::cos(theta);
    const T s = std::sin(theta);
#endif
    const T thInv = T(1) / theta;
    const T w0 = ax * thInv, w1 = ay * thInv, w2 = az * thInv;
    const T wx0 = w1 * X2 - w2 * X1;
    const T wx1 = w2 * X0 - w0 * X2;
    const T wx2 = w0 * X1 - w1 * X0;
    const T wdX = w0 * X0 + w1 * X1 + w2 * X2;
    const T omc = T(1) - c;
    P[0] = c * X0 + s * wx0 + omc * wdX * w0;
    P[1] = c * X1 + s * wx1 + omc * wdX * w1;
    P[2] = c * X2 + s * wx2 + omc * wdX * w2;
    // R = c I + s [w]x + (1-c) w w^T
    const T w[3] = {w0, w1, w2};
    const T wx[3] = {wx0, wx1, wx2};
    const T Xv[3] = {X0, X1, X2};
    R[0][0] = c + omc * w0 * w0;
    R[0][1] = -s * w2 + omc * w0 * w1;
    R[0][2] = s * w1 + omc * w0 * w2;
    R[1][0] = s * w2 + omc * w1 * w0;
    R[1][1] = c + omc * w1 * w1;
    R[1][2] = -s * w0 + omc * w1 * w2;
    R[2][0] = -s * w1 + omc * w2 * w0;
    R[2][1] = s * w0 + omc * w2 * w1;
    R[2][2] = c + omc * w2 * w2;
    // dP/daa = A w^T + M (I - w w^T) / theta,
    // M = s (-[X]x) + (1-c) (w X^T + (w.X) I)
    T A[3], M[3][3];
    for (int i = 0; i < 3; ++i) A[i] = -s * Xv[i] + c * wx[i] + s * wdX * w[i];
    // -[X]x = [[0, X2, -X1], [-X2, 0, X0], [X1, -X0, 0]]
    M[0][0] = omc * (w0 * X0 + wdX);
    M[0][1] = s * X2 + omc * w0 * X1;
    M[0][2] = -s * X1 + omc * w0 * X2;
    M[1][0] = -s * X2 + omc * w1 * X0;
    M[1][1] = omc * (w1 * X1 + wdX);
    M[1][2] = s * X0 + omc * w1 * X2;
    M[2][0] = s * X1 + omc * w2 * X0;
    M[2][1] = -s * X0 + omc * w2 * X1;
    M[2][2] = omc * (w2 * X2 + wdX);
    for (int i = 0; i < 3; ++i) {
      const T Mw = M[i][0] * w0 + M[i][1] * w1 + M[i][2] * w2;
      for (int j = 0; j < 3; ++j)
        dPda[i][j] = A[i] * w[j] + (M[i][j] - Mw * w[j]) * thInv;
    }
  } else {
    // theta -> 0: P = X + aa x X, R = I + [aa]x, dP/daa = -[X]x.
    P[0] = X0 + (ay * X2 - az * X1);
    P[1] = X1 + (az * X0 - ax * X2);
    P[2] = X2 + (ax * X1 - ay * X0);
    R[0][0] = T(1);
    R[0][1] = -az;
    R[0][2] = ay;
    R[1][0] = az;
    R[1][1] = T(1);
    R[1][2] = -ax;
    R[2][0] = -ay;
    R[2][1] = ax;
    R[2][2] = T(1);
    dPda[0][0] = T(0);
    dPda[0][1] = X2;
    dPda[0][2] = -X1;
    dPda[1][0] = -X2;
    dPda[1][1] = T(0);
    dPda[1][2] = X0;
    dPda[2][0] = X1;
    dPda[2][1] = -X0;
    dPda[2][2] = T(0);
  }
  P[0] += cam[3];
  P[1] += cam[4];
  P[2] += cam[5];

  const T iz = T(1) / P[2];
  const T px = -P[0] * iz;
  const T py = -P[1] * iz;
  // dp/dP
  const T dpdP[2][3] = {{-iz, T(0), -px * iz}, {T(0), -iz, -py * iz}};
  const T r2 = px * px + py * py;
  const T f = cam[6], k1 = cam[7], k2 = cam[8];
  const T d = T(1) + r2 * (k1 + k2 * r2);
  res[0] = f * d * px - meas[0];
  res[1] = f * d * py - meas[1];
  // dres/dp = f (d I + 2 (k1 + 2 k2 r2) p p^T)
  const T q = T(2) * (k1 + T(2) * k2 * r2) * f;
  const T drdp[2][2] = {{f * d + q * px * px, q * px * py},
                        {q * px * py, f * d + q * py * py}};
  // dres/dP = dres/dp * dp/dP  (2x3)
  T drdP[2][3];
  for (int r = 0; r < 2; ++r)
    for (int j = 0; j < 3; ++j)
      drdP[r][j] = drdp[r][0] * dpdP[0][j] + drdp[r][1] * dpdP[1][j];
  for (int r = 0; r < 2; ++r) {
    // aa columns
    for (int j = 0; j < 3; ++j)
      Jc[r][j] = drdP[r][0] * dPda[0][j] + drdP[r][1] * dPda[1][j] +
                 drdP[r][2] * dPda[2][j];
    // t columns (dP/dt = I)
    for (int j = 0; j < 3; ++j) Jc[r][3 + j] = drdP[r][j];
    // point columns (dP/dX = R)
    for (int j = 0; j < 3; ++j)
      Jp[r][j] = drdP[r][0] * R[0][j] + drdP[r][1] * R[1][j] +
                 drdP[r][2] * R[2][j];
  }
  const T pvec[2] = {px, py};
  for (int r = 0; r < 2; ++r) {
    Jc[r][6] = d * pvec[r];
    Jc[r][7] = f * r2 * pvec[r];
    Jc[r][8] = f * r2 * r2 * pvec[r];
  }
}

}  // namespace megba
