// Small dense symmetric-positive-definite block helpers (host + device).
// Used for the Schur block-Jacobi preconditioner and the Hll block inverse
// (the reference used cublas<t>matinvBatched; here: in-register Cholesky,
// /root/reference/src/solver/schur_pcg_solver.cu:60-97 is the behavioural
// anchor).
#pragma once

#include "jet.hpp"  // for MEGBA_HD

namespace megba {

// Invert a symmetric positive definite DxD matrix stored row-major (full
// storage) via Cholesky: A = L L^T, then A^{-1} = L^{-T} L^{-1}.
// All in registers/stack; D is a compile-time constant (3 or 9 here).
template <typename T, int D>
MEGBA_HD inline bool spdInvert(const T* a, T* inv) {
  T L[D][D];
  // Cholesky factorization (lower).
  for (int i = 0; i < D; ++i) {
    for (int j = 0; j <= i; ++j) {
      T s = a[i * D + j];
      for (int k = 0; k < j; ++k) s -= L[i][k] * L[j][k];
      if (i == j) {
        if (!(s > T(0))) return false;
#ifdef __HIP_DEVICE_COMPILE__
        L[i][j] = ::sqrt(s);
#else
        L[i][j] = std::sqrt(s);
#endif
      } else {
        L[i][j] = s / L[j][j];
      }
    }
  }
  // Invert L in place (lower triangular): Linv.
  T Linv[D][D];
  for (int i = 0; i < D; ++i) {
    Linv[i][i] = T(1) / L[i][i];
    for (int j = 0; j < i; ++j) {
      T s = T(0);
      for (int k = j; k < i; ++k) s -= L[i][k] * Linv[k][j];
      Linv[i][j] = s / L[i][i];
    }
  }
  // inv = Linv^T * Linv  (symmetric; fill full storage).
  for (int i = 0; i < D; ++i) {
    for (int j = 0; j <= i; ++j) {
      T s = T(0);
      for (int k = i; k < D; ++k) s += Linv[k][i] * Linv[k][j];
      inv[i * D + j] = s;
      inv[j * D + i] = s;
    }
  }
  return true;
}

// Packed-lower in-place SPD inverse: uses D*(D+1)/2 scratch values instead of
// 2*D*D, so a 9x9 fp64 inverse stays in registers on gfx950 (a full-storage
// Cholesky would need ~324 VGPRs and spill to scratch).  Used by the
// block-inverse kernels (thread per block).
template <typename T, int D>
MEGBA_HD inline bool spdInvertPacked(const T* a, T* inv) {
  constexpr int P = D * (D + 1) / 2;
  T L[P];  // packed row-major lower: L[i*(i+1)/2 + j], j<=i
  // Cholesky.
  for (int i = 0, ii = 0; i < D; ii += ++i) {
    for (int j = 0, jj = 0; j <= i; jj += ++j) {
      T s = a[i * D + j];
      for (int k = 0; k < j; ++k) s -= L[ii + k] * L[jj + k];
      if (i == j) {
        if (!(s > T(0))) return false;
#ifdef __HIP_DEVICE_COMPILE__
        L[ii + i] = ::sqrt(s);
#else
        L[ii + i] = std::sqrt(s);
#endif
      } else {
        L[ii + j] = s / L[jj + j];
      }
    }
  }
  // Invert the lower-triangular factor in place.
  for (int i = 0, ii = 0; i < D; ii += ++i) {
    const T dinv = T(1) / L[ii + i];
    L[ii + i] = dinv;
    for (int j = 0; j < i; ++j) {
      T s = T(0);
      // s = -sum_{k=j..i-1} L[i][k] * Linv[k][j]; Linv rows < i already done,
      // and row i's entries at k < j are the only overwritten ones (not read).
      for (int k = j; k < i; ++k) {
        const int krow = k * (k + 1) / 2;
        s -= L[ii + k] * L[krow + j];
      }
      L[ii + j] = s * dinv;
    }
  }
  // inv = Linv^T Linv (full storage output).
  for (int i = 0; i < D; ++i)
    for (int j = 0; j <= i; ++j) {
      T s = T(0);
      for (int k = i; k < D; ++k) {
        const int krow = k * (k + 1) / 2;
        s += L[krow + i] * L[krow + j];
      }
      inv[i * D + j] = s;
      inv[j * D + i] = s;
    }
  return true;
}

// y = A x for a DxD row-major matrix.
template <typename T, int D>
MEGBA_HD inline void matVec(const T* a, const T* x, T* y) {
  for (int i = 0; i < D; ++i) {
    T s = T(0);
    for (int j = 0; j < D; ++j) s += a[i * D + j] * x[j];
    y[i] = s;
  }
}

}  // namespace megba
