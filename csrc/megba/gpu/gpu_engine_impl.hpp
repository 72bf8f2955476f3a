// MI355X (gfx950) engine: HIP kernels + host orchestration + RCCL collectives.
//
// Kernel / distribution design (MI355X-first redesign, not a port —
// behavioural anchors cite /root/reference):
//  * Observations are (point, camera)-sorted and partitioned on POINT
//    boundaries: each rank owns its points outright, so the whole point side
//    (Hll, g_p, Cinv, E^T x, back-substitution) is communication-free and
//    the only per-PCG-iteration collective is an allreduce of the replicated
//    camera vector (9*ncam = 128 KB on Venice).  The reference replicates
//    the point side instead and allreduces 3*npt words (24 MB on Venice)
//    per iteration (its sites A1/A3-A6) — a ~190x traffic reduction sized
//    for xGMI's per-link ring bandwidth.
//  * kForward fuses the ENTIRE vectorised-autodiff forward pass into one
//    kernel: each observation is handled by 4 consecutive lanes, each lane
//    carrying a Jet<T,3> slice of the 12-wide dual part, so every
//    intermediate of the reprojection expression lives in VGPRs.  The
//    reference launches ~1 CUDA kernel per elementary op and streams
//    (N+1)*nItem doubles through HBM each time
//    (src/operator/jet_vector_math_impl.cu).
//  * Assembly transports the camera-side per-edge data through a cam-sorted
//    edge-major slab (contiguous ~400 B per edge, ~1 line-fetch per 64 B)
//    instead of per-element transpose gathers (8x amplification) or
//    per-element atomics (the all-atomic version measured 40.8 ms on
//    Venice-5M, profiles/r01_venice_single_gpu.md).  Point-side Hll/g_p
//    accumulate with a wave-level segmented scan over the point-sorted runs
//    — atomics only at segment tails.
//  * Control-flow scalars (rho, p^T q, norms) use fixed-shape two-pass
//    deterministic reductions so every rank takes identical PCG branches.
//  * r2 additions: every kernel and the engine are templated over the
//    block dims (camDim, ptDim, resDim) — instantiation TUs
//    gpu_dims_*.hip, runtime dispatch in gpu_engine.hip; the per-
//    iteration product kernels read 16-byte packed vector-group J/Hpl
//    layouts with 4-padded w and XP-padded x gathers; the camera-chunk
//    table is point-band-blocked so the gathered w slice stays
//    L2-resident; the PCG body is hipGraph-captured (incl. the RCCL
//    allreduce) with device pointer-slot indirection for the double-
//    buffered J set; fused-vs-separate E^T x + Cinv is auto-tuned per
//    problem.  Measured history: profiles/r02_gather_bands.md.
#pragma once

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cmath>
#include <condition_variable>
#include <cstdlib>
#include <cstring>
#include <memory>
#include <mutex>
#include <thread>
#include <type_traits>
#include <utility>
#include <vector>

#include "../analytical.hpp"
#include "../bal_functor.hpp"
#include "../jv/jetvector.hpp"
#include "../smallmat.hpp"
#include "gpu_engine.hpp"

namespace megba {

#ifndef HIP_CHECK
#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    MEGBA_CHECK(_e == hipSuccess,                                           \
                std::string("HIP error: ") + hipGetErrorString(_e) + " @ " #expr); \
  } while (0)
#endif

#ifndef RCCL_CHECK
#define RCCL_CHECK(expr)                                                    \
  do {                                                                      \
    ncclResult_t _e = (expr);                                               \
    MEGBA_CHECK(_e == ncclSuccess,                                          \
                std::string("RCCL error: ") + ncclGetErrorString(_e) + " @ " #expr); \
  } while (0)
#endif

namespace {

constexpr int kBlk = 256;
constexpr int kRedBlocks = 256;  // fixed -> deterministic reductions

inline int gridFor(int64_t n) {
  int64_t g = (n + kBlk - 1) / kBlk;
  return (int)(g < 1 ? 1 : (g > 8192 ? 8192 : g));
}

// ---------------------------------------------------------------------------
// Reductions
// ---------------------------------------------------------------------------
enum class ROp { Dot, SumSq, AbsMax };

// 16-byte vector groups used by the packed J / Hpl layouts.
template <typename T>
struct PackVec {
  static constexpr int VEC = 16 / sizeof(T);
  typedef T type __attribute__((ext_vector_type(16 / sizeof(T))));
};

template <typename T, ROp OP>
__global__ void kRedPartial(const T* a, const T* b, int64_t n, double* part) {
  __shared__ double sm[kBlk];
  double v = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk) {
    const double x = (double)a[i];
    if (OP == ROp::Dot)
      v += x * (double)b[i];
    else if (OP == ROp::SumSq)
      v += x * x;
    else
      v = fmax(v, fabs(x));
  }
  sm[threadIdx.x] = v;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
      if (OP == ROp::AbsMax)
        sm[threadIdx.x] = fmax(sm[threadIdx.x], sm[threadIdx.x + s]);
      else
        sm[threadIdx.x] += sm[threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) part[blockIdx.x] = sm[0];
}

template <ROp OP>
__global__ void kRedFinal(const double* part, int nb, double* out) {
  __shared__ double sm[kBlk];
  double v = 0.0;
  for (int i = threadIdx.x; i < nb; i += kBlk) {
    if (OP == ROp::AbsMax)
      v = fmax(v, part[i]);
    else
      v += part[i];
  }
  sm[threadIdx.x] = v;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
      if (OP == ROp::AbsMax)
        sm[threadIdx.x] = fmax(sm[threadIdx.x], sm[threadIdx.x + s]);
      else
        sm[threadIdx.x] += sm[threadIdx.x + s];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) *out = sm[0];
}

// ---------------------------------------------------------------------------
// Forward (fused register autodiff)
// ---------------------------------------------------------------------------
// Generic fused-autodiff forward: each observation is handled by
// LANES = ceil((CD+PD)/3) consecutive work items, each carrying a Jet<T,3>
// slice of the CD+PD-wide dual part.
template <typename T, int CD, int PD, int RD>
__global__ __launch_bounds__(256, 2) void kForward(
    int64_t nL, const int* __restrict__ camOf, const int* __restrict__ ptOf,
    const T* __restrict__ params, int ncam, const T* __restrict__ meas,
    const unsigned char* __restrict__ camFixed,
    const unsigned char* __restrict__ ptFixed, T* __restrict__ rOut,
    T* __restrict__ Jc, T* __restrict__ Jp, double* chi2Acc, int lossKind,
    T lossD2, const T* __restrict__ intr3) {
  using J3 = Jet<T, 3>;
  constexpr int GW = CD + PD;
  constexpr int LANES = (GW + 2) / 3;
  __shared__ double sm[kBlk];
  double chi2 = 0.0;
  const T* ptsBase = params + (int64_t)ncam * CD;
  const int64_t nWork = nL * LANES;
  T intr[3] = {intr3[0], intr3[1], intr3[2]};
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < nWork;
       i += (int64_t)gridDim.x * kBlk) {
    const int64_t e = i / LANES;
    const int sub = (int)(i % LANES);   // this lane's 3-wide gradient slice
    const int base = 3 * sub;           // gradient columns [base,base+3)
    const T* cp = params + (int64_t)camOf[e] * CD;
    const T* pp = ptsBase + (int64_t)ptOf[e] * PD;
    J3 cam[CD], pt[PD], res[RD];
    for (int k = 0; k < CD; ++k) cam[k] = J3::leaf(cp[k], k - base);
    for (int k = 0; k < PD; ++k) pt[k] = J3::leaf(pp[k], CD + k - base);
    T m[RD];
    for (int k = 0; k < RD; ++k) m[k] = meas[RD * e + k];
    builtinResidual<T, J3, CD, PD, RD>(cam, pt, m, intr, res);
    const bool cfix = camFixed && camFixed[camOf[e]];
    const bool pfix = ptFixed && ptFixed[ptOf[e]];
    for (int row = 0; row < RD; ++row) {
      for (int j = 0; j < 3; ++j) {
        const int col = base + j;
        if (col >= GW) break;
        if (col < CD)
          Jc[((int64_t)(col * RD + row)) * nL + e] =
              cfix ? T(0) : res[row].d[j];
        else
          Jp[((int64_t)((col - CD) * RD + row)) * nL + e] =
              pfix ? T(0) : res[row].d[j];
      }
      if (sub == 0) rOut[(int64_t)row * nL + e] = res[row].v;
    }
    if (sub == 0) {
      double ss = 0.0;
      for (int row = 0; row < RD; ++row)
        ss += (double)res[row].v * (double)res[row].v;
      chi2 += (double)lossRho(lossKind, lossD2, (T)ss);
    }
  }
  sm[threadIdx.x] = chi2;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) sm[threadIdx.x] += sm[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(chi2Acc, sm[0]);
}

// Value-share forward experiment (env MEGBA_FWD_VS=1,
// BAL (9,3,2) only): the 4 gradient lanes of an edge recompute the whole
// VALUE chain of the reprojection; the dual math needs every intermediate
// value on every lane, so only the EXPENSIVE scalar values (sqrt, sincos,
// two reciprocals — the transcendental tail of the chain) are computed
// once on the edge's leader lane and broadcast with __shfl; their jets
// are then assembled from the analytic derivative rules.  Cheap mul/add
// values stay redundantly computed (a shuffle costs as much as an FMA).
template <typename T>
__global__ __launch_bounds__(256, 2) void kForwardVS(
    int64_t nL, const int* __restrict__ camOf, const int* __restrict__ ptOf,
    const T* __restrict__ params, int ncam, const T* __restrict__ meas,
    const unsigned char* __restrict__ camFixed,
    const unsigned char* __restrict__ ptFixed, T* __restrict__ rOut,
    T* __restrict__ Jc, T* __restrict__ Jp, double* chi2Acc, int lossKind,
    T lossD2) {
  using J3 = Jet<T, 3>;
  __shared__ double sm[kBlk];
  double chi2 = 0.0;
  const T* ptsBase = params + (int64_t)ncam * 9;
  const int64_t nWork = nL * 4;
  const int lane = (int)threadIdx.x & 63;
  const int leader = lane & ~3;
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < nWork;
       i += (int64_t)gridDim.x * kBlk) {
    const int64_t e = i >> 2;
    const int sub = (int)(i & 3);
    const int base = 3 * sub;
    const T* cp = params + (int64_t)camOf[e] * 9;
    const T* pp = ptsBase + (int64_t)ptOf[e] * 3;
    J3 cam[9], pt[3], res[2];
    for (int k = 0; k < 9; ++k) cam[k] = J3::leaf(cp[k], k - base);
    for (int k = 0; k < 3; ++k) pt[k] = J3::leaf(pp[k], 9 + k - base);
    const T m[2] = {meas[2 * e], meas[2 * e + 1]};
    // ---- Rodrigues with shared transcendental values ----
    const J3 theta2 =
        cam[0] * cam[0] + cam[1] * cam[1] + cam[2] * cam[2];
    J3 P[3];
    if (theta2.v > T(1e-14)) {  // edge-uniform branch
      T thv = T(0), stv = T(0), ctv = T(0), thInvV = T(0);
      if (sub == 0) {
        thv = ::sqrt(theta2.v);
        stv = ::sin(thv);
        ctv = ::cos(thv);
        thInvV = T(1) / thv;
      }
      stv = __shfl(stv, leader, 64);
      ctv = __shfl(ctv, leader, 64);
      thInvV = __shfl(thInvV, leader, 64);
      // jets from the analytic rules: d(sqrt) = d(theta2)/(2 theta) etc.
      J3 theta, sinth, costh, thetaInv;
      theta.v = thv == T(0) ? (thv = T(1) / thInvV) : thv;  // non-leader thv
      const T half = T(0.5) * thInvV;
      for (int k = 0; k < 3; ++k) theta.d[k] = theta2.d[k] * half;
      sinth.v = stv;
      costh.v = ctv;
      thetaInv.v = thInvV;
      const T mInv2 = -thInvV * thInvV;
      for (int k = 0; k < 3; ++k) {
        sinth.d[k] = ctv * theta.d[k];
        costh.d[k] = -stv * theta.d[k];
        thetaInv.d[k] = mInv2 * theta.d[k];
      }
      const J3 w0 = cam[0] * thetaInv;
      const J3 w1 = cam[1] * thetaInv;
      const J3 w2 = cam[2] * thetaInv;
      const J3 wxp0 = w1 * pt[2] - w2 * pt[1];
      const J3 wxp1 = w2 * pt[0] - w0 * pt[2];
      const J3 wxp2 = w0 * pt[1] - w1 * pt[0];
      const J3 wdp =
          (w0 * pt[0] + w1 * pt[1] + w2 * pt[2]) * (T(1) - costh);
      P[0] = pt[0] * costh + wxp0 * sinth + w0 * wdp;
      P[1] = pt[1] * costh + wxp1 * sinth + w1 * wdp;
      P[2] = pt[2] * costh + wxp2 * sinth + w2 * wdp;
    } else {
      P[0] = pt[0] + (cam[1] * pt[2] - cam[2] * pt[1]);
      P[1] = pt[1] + (cam[2] * pt[0] - cam[0] * pt[2]);
      P[2] = pt[2] + (cam[0] * pt[1] - cam[1] * pt[0]);
    }
    P[0] += cam[3];
    P[1] += cam[4];
    P[2] += cam[5];
    // shared reciprocal of the depth
    T invNegZv = T(0);
    if (sub == 0) invNegZv = T(-1) / P[2].v;
    invNegZv = __shfl(invNegZv, leader, 64);
    J3 invNegZ;
    invNegZ.v = invNegZv;
    const T q = invNegZv * invNegZv;  // d(-1/z) = z' / z^2
    for (int k = 0; k < 3; ++k) invNegZ.d[k] = q * P[2].d[k];
    const J3 xp = P[0] * invNegZ;
    const J3 yp = P[1] * invNegZ;
    const J3 r2 = xp * xp + yp * yp;
    const J3 distortion = T(1) + r2 * (cam[7] + cam[8] * r2);
    const J3 scaled = cam[6] * distortion;
    res[0] = scaled * xp - m[0];
    res[1] = scaled * yp - m[1];
    // ---- identical epilogue to kForward ----
    const bool cfix = camFixed && camFixed[camOf[e]];
    const bool pfix = ptFixed && ptFixed[ptOf[e]];
    for (int row = 0; row < 2; ++row) {
      for (int j = 0; j < 3; ++j) {
        const int col = base + j;
        if (col < 9)
          Jc[((int64_t)(col * 2 + row)) * nL + e] =
              cfix ? T(0) : res[row].d[j];
        else
          Jp[((int64_t)((col - 9) * 2 + row)) * nL + e] =
              pfix ? T(0) : res[row].d[j];
      }
      if (sub == 0) rOut[(int64_t)row * nL + e] = res[row].v;
    }
    if (sub == 0)
      chi2 += (double)lossRho(lossKind, lossD2,
                              res[0].v * res[0].v + res[1].v * res[1].v);
  }
  sm[threadIdx.x] = chi2;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) sm[threadIdx.x] += sm[threadIdx.x + st];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(chi2Acc, sm[0]);
}

// Analytical-derivative forward: one thread per edge, closed-form residual +
// 2x12 Jacobian (reference C11, src/geo/analytical_derivatives.cu).
// BAL (9,3,2) only; the engine refuses analytical mode for other dims.
template <typename T>
__global__ void kForwardAnalytical(int64_t nL, const int* __restrict__ camOf,
                                   const int* __restrict__ ptOf,
                                   const T* __restrict__ params, int ncam,
                                   const T* __restrict__ meas,
                                   const unsigned char* __restrict__ camFixed,
                                   const unsigned char* __restrict__ ptFixed,
                                   T* __restrict__ rOut, T* __restrict__ Jc,
                                   T* __restrict__ Jp, double* chi2Acc,
                                   int lossKind, T lossD2) {
  __shared__ double sm[kBlk];
  double chi2 = 0.0;
  const T* ptsBase = params + (int64_t)ncam * 9;
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    const T* cp = params + (int64_t)camOf[e] * 9;
    const T* pp = ptsBase + (int64_t)ptOf[e] * 3;
    const T m[2] = {meas[2 * e], meas[2 * e + 1]};
    T res[2], jc[2][9], jp[2][3];
    balAnalytical<T>(cp, pp, m, res, jc, jp);
    chi2 += (double)lossRho(lossKind, lossD2,
                            res[0] * res[0] + res[1] * res[1]);
    const bool cfix = camFixed && camFixed[camOf[e]];
    const bool pfix = ptFixed && ptFixed[ptOf[e]];
    for (int row = 0; row < 2; ++row) {
      rOut[(int64_t)row * nL + e] = res[row];
      for (int col = 0; col < 9; ++col)
        Jc[((int64_t)(col * 2 + row)) * nL + e] = cfix ? T(0) : jc[row][col];
      for (int col = 0; col < 3; ++col)
        Jp[((int64_t)(col * 2 + row)) * nL + e] = pfix ? T(0) : jp[row][col];
    }
  }
  sm[threadIdx.x] = chi2;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) sm[threadIdx.x] += sm[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(chi2Acc, sm[0]);
}

// chi2 with a robust loss, from the residual buffers (custom-edge path).
template <typename T, int RD>
__global__ void kChi2Loss(int64_t nL, const T* __restrict__ r, int lossKind,
                          T lossD2, double* acc) {
  __shared__ double sm[kBlk];
  double local = 0.0;
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    T ss = T(0);
    for (int row = 0; row < RD; ++row) {
      const T rv = r[(int64_t)row * nL + e];
      ss += rv * rv;
    }
    local += (double)lossRho(lossKind, lossD2, ss);
  }
  sm[threadIdx.x] = local;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) sm[threadIdx.x] += sm[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(acc, sm[0]);
}

// Custom-edge support: gather parameter leaves, repack residual JetVectors.
template <typename T, int CD, int PD>
__global__ void kGatherLeaves(int64_t nL, const int* __restrict__ camOf,
                              const int* __restrict__ ptOf,
                              const T* __restrict__ params, int ncam,
                              T* __restrict__ leaf /*[CD+PD][nL]*/) {
  const T* ptsBase = params + (int64_t)ncam * CD;
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    const T* cp = params + (int64_t)camOf[e] * CD;
    const T* pp = ptsBase + (int64_t)ptOf[e] * PD;
    for (int k = 0; k < CD; ++k) leaf[(int64_t)k * nL + e] = cp[k];
    for (int k = 0; k < PD; ++k) leaf[(int64_t)(CD + k) * nL + e] = pp[k];
  }
}

template <typename T, int CD, int PD, int RD>
__global__ void kRepackRes(int64_t nL, int row, const T* __restrict__ rv,
                           const T* __restrict__ rg /*[CD+PD][nL]*/,
                           const int* __restrict__ camOf,
                           const int* __restrict__ ptOf,
                           const unsigned char* __restrict__ camFixed,
                           const unsigned char* __restrict__ ptFixed,
                           T* __restrict__ rOut, T* __restrict__ Jc,
                           T* __restrict__ Jp) {
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    rOut[(int64_t)row * nL + e] = rv[e];
    const bool cfix = camFixed && camFixed[camOf[e]];
    const bool pfix = ptFixed && ptFixed[ptOf[e]];
    for (int k = 0; k < CD; ++k)
      Jc[((int64_t)(k * RD + row)) * nL + e] =
          cfix ? T(0) : rg[(int64_t)k * nL + e];
    for (int k = 0; k < PD; ++k)
      Jp[((int64_t)(k * RD + row)) * nL + e] =
          pfix ? T(0) : rg[(int64_t)(CD + k) * nL + e];
  }
}

// ---------------------------------------------------------------------------
// Assembly
// ---------------------------------------------------------------------------
// Slab row layout (T values), written at the edge's CAM-sorted position:
//   [0..CD*PD)                    Hpl block (EXPL only)
//   [JCOFF..JCOFF+CD*RD)          Jc rows (col-major groups: jc[col][row])
//   [WROFF..WROFF+RD)             weighted residual rows
//   [WJCOFF..WJCOFF+CD*RD)        weighted Jc rows (HASINFO only)
//   [JPOFF..JPOFF+PD*RD)          Jp rows (implicit only)
template <int CD, int PD, int RD, bool EXPL, bool HASINFO>
struct SlabLayout {
  static constexpr int JCOFF = EXPL ? CD * PD : 0;
  static constexpr int WROFF = JCOFF + CD * RD;
  static constexpr int WJCOFF = WROFF + RD;
  static constexpr int JPOFF = WJCOFF + (HASINFO ? CD * RD : 0);
  static constexpr int SW = JPOFF + (EXPL ? 0 : PD * RD);
};

// Packed-upper symmetric index for an RDxRD (or PDxPD) matrix.
template <int D>
__device__ __host__ constexpr int symIdx(int i, int j) {
  return i <= j ? i * D - i * (i - 1) / 2 + (j - i)
                : j * D - j * (j - 1) / 2 + (i - j);
}

// Per-edge pass, primary ((pt,cam)-sorted) order: Hpl grad-major (for E^T x),
// the cam-sorted slab row, and the point-side Hll/g_p via a wave-level
// segmented scan (atomics only at point-run tails).
template <typename T, int CD, int PD, int RD, bool HASINFO, bool EXPL>
__global__ void kAssembleEdge(int64_t nL, const int* __restrict__ camOf,
                              const int* __restrict__ ptOf,
                              const T* __restrict__ r, const T* __restrict__ Jc,
                              const T* __restrict__ Jp,
                              const T* __restrict__ info, T* __restrict__ Hll,
                              T* __restrict__ Hpl, T* __restrict__ g, int ncam,
                              const int* __restrict__ camPos,
                              T* __restrict__ slab, int lossKind, T lossD2,
                              T* __restrict__ jPk) {
  using L = SlabLayout<CD, PD, RD, EXPL, HASINFO>;
  constexpr int RW = RD * (RD + 1) / 2;   // packed info entries
  constexpr int PH = PD * (PD + 1) / 2;   // packed Hll entries
  T* gp = g + (int64_t)ncam * CD;
  const int lane = threadIdx.x & 63;
  const int64_t nWork = ((nL + kBlk - 1) / kBlk) * (int64_t)kBlk;
  for (int64_t e0i = blockIdx.x * (int64_t)kBlk + threadIdx.x; e0i < nWork;
       e0i += (int64_t)gridDim.x * kBlk) {
    const bool active = e0i < nL;
    const int64_t e = active ? e0i : nL - 1;
    const int pt = ptOf[e];
    T hllP[PH], gpP[PD];
    for (int k = 0; k < PH; ++k) hllP[k] = T(0);
    for (int k = 0; k < PD; ++k) gpP[k] = T(0);
    if (active) {
      T jc[RD][CD], jp[RD][PD], rr[RD];
      for (int k = 0; k < CD; ++k)
        for (int row = 0; row < RD; ++row)
          jc[row][k] = Jc[((int64_t)(k * RD + row)) * nL + e];
      for (int k = 0; k < PD; ++k)
        for (int row = 0; row < RD; ++row)
          jp[row][k] = Jp[((int64_t)(k * RD + row)) * nL + e];
      for (int row = 0; row < RD; ++row) rr[row] = r[(int64_t)row * nL + e];
      T wjp[RD][PD], wr[RD];
      T wjc[RD][CD];
      if (HASINFO) {
        // HASINFO also covers robust-loss runs without an information
        // matrix (info == nullptr -> identity), since both need the
        // weighted Jc rows stored separately in the slab.
        T W[RW];
        if (info) {
          for (int k = 0; k < RW; ++k) W[k] = info[RW * e + k];
        } else {
          for (int k = 0; k < RW; ++k) W[k] = T(0);
          for (int i = 0; i < RD; ++i) W[symIdx<RD>(i, i)] = T(1);
        }
        if (lossKind) {
          T ss = T(0);
          for (int row = 0; row < RD; ++row) ss += rr[row] * rr[row];
          const T w = lossWeight(lossKind, lossD2, ss);
          for (int k = 0; k < RW; ++k) W[k] *= w;
        }
        for (int i = 0; i < RD; ++i) {
          for (int k = 0; k < CD; ++k) {
            T v = T(0);
            for (int j = 0; j < RD; ++j) v += W[symIdx<RD>(i, j)] * jc[j][k];
            wjc[i][k] = v;
          }
          for (int k = 0; k < PD; ++k) {
            T v = T(0);
            for (int j = 0; j < RD; ++j) v += W[symIdx<RD>(i, j)] * jp[j][k];
            wjp[i][k] = v;
          }
          T v = T(0);
          for (int j = 0; j < RD; ++j) v += W[symIdx<RD>(i, j)] * rr[j];
          wr[i] = v;
        }
      } else {
        for (int i = 0; i < RD; ++i) {
          for (int k = 0; k < PD; ++k) wjp[i][k] = jp[i][k];
          wr[i] = rr[i];
        }
      }
      T* row = slab + (int64_t)camPos[e] * L::SW;
      if (EXPL) {
        // Hpl stored as 16-byte vector groups ([group][nL][VEC]) so the
        // per-iteration E^T x / E w reads are dwordx4 loads (r2, same
        // treatment as the implicit packed J — profiles/r02_gather_bands.md)
        constexpr int VEC = 16 / (int)sizeof(T);
        constexpr int NGH = (CD * PD + VEC - 1) / VEC;
        T v[NGH * VEC];
        for (int k = 0; k < NGH * VEC; ++k) v[k] = T(0);
        for (int a = 0; a < CD; ++a)
          for (int b = 0; b < PD; ++b) {
            T acc = T(0);
            for (int i = 0; i < RD; ++i) acc += jc[i][a] * wjp[i][b];
            v[a * PD + b] = acc;
            row[a * PD + b] = acc;
          }
        using TV = typename PackVec<T>::type;
        TV* o = (TV*)Hpl;
        for (int g = 0; g < NGH; ++g) {
          TV vv;
          for (int q = 0; q < VEC; ++q) vv[q] = v[g * VEC + q];
          o[(int64_t)g * nL + e] = vv;
        }
      }
      for (int k = 0; k < CD; ++k)
        for (int i = 0; i < RD; ++i) row[L::JCOFF + k * RD + i] = jc[i][k];
      for (int i = 0; i < RD; ++i) row[L::WROFF + i] = wr[i];
      if (HASINFO)
        for (int k = 0; k < CD; ++k)
          for (int i = 0; i < RD; ++i) row[L::WJCOFF + k * RD + i] = wjc[i][k];
      if (!EXPL) {
        for (int k = 0; k < PD; ++k)
          for (int i = 0; i < RD; ++i) row[L::JPOFF + k * RD + i] = jp[i][k];
        // primary-order packed [Jc, Jp] copy for the per-iteration E^T x
        // reads — written here from registers (the J values are already
        // loaded), replacing a separate full-pass pack kernel.
        {
          using TV = typename PackVec<T>::type;
          constexpr int VEC = PackVec<T>::VEC;
          constexpr int CR = CD * RD, PR = PD * RD;
          constexpr int NGJ = (CR + PR + VEC - 1) / VEC;
          TV* o = (TV*)jPk;
          for (int gI = 0; gI < NGJ; ++gI) {
            TV v;
            for (int q = 0; q < VEC; ++q) {
              const int k = gI * VEC + q;
              v[q] = k < CR ? jc[k % RD][k / RD]
                            : (k < CR + PR ? jp[(k - CR) % RD][(k - CR) / RD]
                                           : T(0));
            }
            o[(int64_t)gI * nL + e] = v;
          }
        }
      }
      // point-side contributions (packed upper)
      for (int a = 0; a < PD; ++a)
        for (int b = a; b < PD; ++b) {
          T v = T(0);
          for (int i = 0; i < RD; ++i) v += jp[i][a] * wjp[i][b];
          hllP[symIdx<PD>(a, b)] = v;
        }
      for (int k = 0; k < PD; ++k) {
        T v = T(0);
        for (int i = 0; i < RD; ++i) v += jp[i][k] * wr[i];
        gpP[k] = -v;
      }
    }
    // segmented scan over the wave's point runs (early exit once no lane
    // continues a segment at this distance — monotone in off)
    for (int off = 1; off < 64; off <<= 1) {
      const int ppt = __shfl_up(pt, off, 64);
      const bool join = lane >= off && ppt == pt;
      if (__ballot(join) == 0ull) break;
      T aH[PH], aG[PD];
      for (int k = 0; k < PH; ++k) aH[k] = __shfl_up(hllP[k], off, 64);
      for (int k = 0; k < PD; ++k) aG[k] = __shfl_up(gpP[k], off, 64);
      if (join) {
        for (int k = 0; k < PH; ++k) hllP[k] += aH[k];
        for (int k = 0; k < PD; ++k) gpP[k] += aG[k];
      }
    }
    const int nextPt = __shfl_down(pt, 1, 64);
    const bool tail = active && (lane == 63 || nextPt != pt || e0i == nL - 1);
    if (tail) {
      T* H = Hll + (int64_t)pt * PD * PD;
      for (int i = 0; i < PD; ++i)
        for (int j = 0; j < PD; ++j)
          atomicAdd(&H[i * PD + j], hllP[symIdx<PD>(i, j)]);
      for (int k = 0; k < PD; ++k)
        atomicAdd(&gp[(int64_t)pt * PD + k], gpP[k]);
    }
  }
}

// Camera blocks: one 128-thread block per <=256-row chunk of one camera's
// cam-sorted slab rows.  Rows are contiguous, staged through LDS in 128-row
// tiles; threads 0..CD*CD-1 each own one Hpp element, the next CD one g_c
// element; one atomicAdd per output per chunk.
template <typename T, int CD, int PD, int RD, bool HASINFO, bool EXPL>
__global__ __launch_bounds__(128) void kAssembleCam(
    int nChunks, const int* __restrict__ chCam, const int* __restrict__ chLo,
    const int* __restrict__ chHi, const T* __restrict__ slab,
    T* __restrict__ Hpp, T* __restrict__ g) {
  using L = SlabLayout<CD, PD, RD, EXPL, HASINFO>;
  constexpr int CC = CD * CD;
  constexpr int CR = CD * RD;
  constexpr int TE = 128;
  constexpr int ST = CR + RD + (HASINFO ? CR : 0);  // jc + wr (+ wjc)
  __shared__ T lds[TE * ST];
  const int chunk = blockIdx.x;
  if (chunk >= nChunks) return;
  const int cam = chCam[chunk];
  const int lo = chLo[chunk], hi = chHi[chunk];
  const int t = threadIdx.x;
  T acc = T(0);
  const int ti = t < CC ? t / CD : t - CC;
  const int tj = t < CC ? t % CD : 0;
  for (int s0 = lo; s0 < hi; s0 += TE) {
    const int nt = min(TE, hi - s0);
    for (int idx = t; idx < nt * ST; idx += 128) {
      const int row = idx / ST;
      const int k = idx % ST;
      const T* src = slab + (int64_t)(s0 + row) * L::SW + L::JCOFF;
      lds[row * ST + k] = src[k];
    }
    __syncthreads();
    if (t < CC + CD) {
      const int woff = HASINFO ? CR + RD : 0;  // weighted rows (== raw w/o)
      for (int e = 0; e < nt; ++e) {
        const T* row = lds + e * ST;
        if (t < CC) {
          T v = T(0);
          for (int i = 0; i < RD; ++i)
            v += row[ti * RD + i] * row[woff + tj * RD + i];
          acc += v;
        } else {
          T v = T(0);
          for (int i = 0; i < RD; ++i) v += row[ti * RD + i] * row[CR + i];
          acc -= v;
        }
      }
    }
    __syncthreads();
  }
  if (t < CC)
    atomicAdd(&Hpp[(int64_t)cam * CC + ti * CD + tj], acc);
  else if (t < CC + CD)
    atomicAdd(&g[(int64_t)cam * CD + ti], acc);
}

// MFMA experiment (reference anchor
// build_linear_system.cu:88-146): the per-chunk Hpp accumulation
// Hpp += Jc^T (W Jc) is a K-reduction GEMM with M=N=9, K=2*rows.  One
// v_mfma_f64_16x16x4_f64 per 4 K-rows computes the whole 16x16 tile
// (rows 9..15 zero-padded); the g_c = -Jc^T wr partial rides along as a
// free 10th B column.  Env-gated (MEGBA_MFMA=1), fp64 BAL dims only.
// Four interleaved accumulators break the dependent-MFMA latency chain.
typedef double megba_d4 __attribute__((ext_vector_type(4)));
template <typename T, int CD, int PD, int RD, bool HASINFO, bool EXPL>
__global__ __launch_bounds__(64) void kAssembleCamMfma(
    int nChunks, const int* __restrict__ chCam, const int* __restrict__ chLo,
    const int* __restrict__ chHi, const T* __restrict__ slab,
    T* __restrict__ Hpp, T* __restrict__ g) {
  static_assert(std::is_same<T, double>::value && CD == 9 && PD == 3 &&
                    RD == 2,
                "MFMA assembly path is fp64 BAL (9,3,2) only");
  using L = SlabLayout<CD, PD, RD, EXPL, HASINFO>;
  constexpr int CR = CD * RD;                       // 18
  constexpr int ST = CR + RD + (HASINFO ? CR : 0);  // jc + wr (+ wjc)
  constexpr int TE = 128;
  __shared__ T lds[TE * ST];
  const int chunk = blockIdx.x;
  if (chunk >= nChunks) return;
  const int cam = chCam[chunk];
  const int lo = chLo[chunk], hi = chHi[chunk];
  const int l = (int)threadIdx.x;
  const int i = l & 15;        // A row (camera col) / D col
  const int kq = l >> 4;       // K quarter within each MFMA's K=4
  constexpr int woff = HASINFO ? CR + RD : 0;
  megba_d4 acc[4];
  for (int a = 0; a < 4; ++a) acc[a] = megba_d4{0.0, 0.0, 0.0, 0.0};
  for (int s0 = lo; s0 < hi; s0 += TE) {
    const int nt = min(TE, hi - s0);
    for (int idx = l; idx < nt * ST; idx += 64) {
      const int row = idx / ST;
      const int k = idx % ST;
      lds[row * ST + k] = slab[(int64_t)(s0 + row) * L::SW + L::JCOFF + k];
    }
    __syncthreads();
    const int K = nt * RD;  // edge-rows in this tile
    for (int k0 = 0; k0 < K; k0 += 16) {
      // 4 MFMAs per iteration, one per accumulator -> 4-deep independence
      for (int a = 0; a < 4; ++a) {
        const int kr = k0 + 4 * a + kq;  // this lane's K index
        T av = T(0), bv = T(0);
        if (kr < K) {
          const int e = kr >> 1;        // RD == 2
          const int r = kr & 1;
          const T* row = lds + e * ST;
          if (i < 9) {
            av = row[i * 2 + r];          // Jc[k][i]
            bv = row[woff + i * 2 + r];   // (W Jc)[k][j=i]
          } else if (i == 9) {
            bv = row[CR + r];             // wr[k] -> g column
          }
        }
        acc[a] = __builtin_amdgcn_mfma_f64_16x16x4f64(av, bv, acc[a], 0, 0,
                                                      0);
      }
    }
    __syncthreads();
  }
  for (int a = 1; a < 4; ++a) acc[0] += acc[a];
  // C/D map for v_mfma_f64_16x16x4_f64 (probed on hardware,
  // tools/mfma_probe.hip): col = lane&15, row = 4*reg + (lane>>4).
  for (int v = 0; v < 4; ++v) {
    const int row = v * 4 + kq;
    const int col = i;
    if (row < 9) {
      if (col < 9)
        atomicAdd(&Hpp[(int64_t)cam * 81 + row * 9 + col], acc[0][v]);
      else if (col == 9)
        atomicAdd(&g[(int64_t)cam * 9 + row], -acc[0][v]);
    }
  }
}

// Explicit only: stream the slab's Hpl blocks out as a cam-sorted grad-major
// copy for the E*w kernel (contiguous reads, coalesced writes).
template <typename T, int CD, int PD, int RD, bool HASINFO>
__global__ void kFinalizeCam(int64_t nL, const T* __restrict__ slab,
                             T* __restrict__ HplCam) {
  using L = SlabLayout<CD, PD, RD, true, HASINFO>;
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int NGH = (CD * PD + VEC - 1) / VEC;
  TV* o = (TV*)HplCam;
  for (int64_t j = blockIdx.x * (int64_t)kBlk + threadIdx.x; j < nL;
       j += (int64_t)gridDim.x * kBlk) {
    const T* row = slab + j * L::SW;
    for (int g = 0; g < NGH; ++g) {
      TV v;
      for (int q = 0; q < VEC; ++q) {
        const int k = g * VEC + q;
        v[q] = k < CD * PD ? row[k] : T(0);
      }
      o[(int64_t)g * nL + j] = v;
    }
  }
}

// (The r01 grad-major implicit copies kFinalizeCamImp / kSpmvExImpG were
// replaced by the packed vector-group kernels below in r2 —
// profiles/r02_gather_bands.md.)

// ---------------------------------------------------------------------------
// Packed-J layout for the per-iteration implicit products
// ---------------------------------------------------------------------------
// The grad-major J layout ([k][nL], one 4/8-byte load per value) makes the
// two matrix-free PCG kernels instruction-ISSUE-bound: ~24 scalar loads
// per edge dominate the ~80-instruction inner loop (measured ~2x the byte
// floor on final13682-fp32, profiles/r01_final13682_implicit_fp32.md).
// Repacking the 24 per-edge J values into 16-byte vector groups
// ([group][nL][VEC], VEC = 16B/sizeof(T)) turns them into NG=24/VEC
// dwordx4 loads (6 for fp32, 12 for fp64), still fully coalesced across
// the 64 consecutive-edge lanes.  The pack runs once per ACCEPTED LM step
// (its cost amortizes over the ~100 PCG iterations that read it); the
// packed buffers are single-buffered so the captured PCG graph needs no
// pointer indirection for them.
// (The standalone kPackJPrimary pass was folded into kAssembleEdge's
// register writes in r2.)

// Packed E^T x: identical math to kSpmvEtx<IMP>, vector-group loads.
template <typename T, int CD, int PD, int RD, bool HASINFO>
__global__ void kSpmvEtxPk(int64_t nL, const int* __restrict__ camOf,
                           const int* __restrict__ ptOf,
                           const T* __restrict__ Jpk,
                           const T* const* __restrict__ jSlots,
                           const T* __restrict__ info, int lossKind, T lossD2,
                           const T* __restrict__ xPad,
                           T* __restrict__ out) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int RW = RD * (RD + 1) / 2;
  constexpr int CR = CD * RD, PR = PD * RD;
  constexpr int NG = (CR + PR + VEC - 1) / VEC;
  const T* rBak = jSlots[2];
  const int lane = threadIdx.x & 63;
  const int64_t nWork = ((nL + kBlk - 1) / kBlk) * (int64_t)kBlk;
  for (int64_t j0 = blockIdx.x * (int64_t)kBlk + threadIdx.x; j0 < nWork;
       j0 += (int64_t)gridDim.x * kBlk) {
    const bool active = j0 < nL;
    const int64_t j = active ? j0 : nL - 1;
    const int pt = ptOf[j];
    T o[PD];
    for (int k = 0; k < PD; ++k) o[k] = T(0);
    if (active) {
      constexpr int XP = (CD + VEC - 1) / VEC * VEC;
      const TV* xv4 = (const TV*)(xPad + (int64_t)camOf[j] * XP);
      TV xbuf[XP / VEC];
#pragma unroll
      for (int l = 0; l < XP / VEC; ++l) xbuf[l] = xv4[l];
      TV buf[NG];
      const TV* src = (const TV*)Jpk;
#pragma unroll
      for (int g = 0; g < NG; ++g) buf[g] = src[(int64_t)g * nL + j];
      T u[RD];
#pragma unroll
      for (int rr = 0; rr < RD; ++rr) {
        T v = T(0);
#pragma unroll
        for (int i = 0; i < CD; ++i) {
          const int k = i * RD + rr;
          v += buf[k / VEC][k % VEC] * xbuf[i / VEC][i % VEC];
        }
        u[rr] = v;
      }
      if (HASINFO) {
        T wu[RD];
        for (int i = 0; i < RD; ++i) {
          T v = T(0);
          for (int k = 0; k < RD; ++k)
            v += info[RW * j + symIdx<RD>(i, k)] * u[k];
          wu[i] = v;
        }
        for (int i = 0; i < RD; ++i) u[i] = wu[i];
      }
      if (lossKind) {
        T ss = T(0);
        for (int rr = 0; rr < RD; ++rr) {
          const T rv = rBak[(int64_t)rr * nL + j];
          ss += rv * rv;
        }
        const T w = lossWeight(lossKind, lossD2, ss);
        for (int rr = 0; rr < RD; ++rr) u[rr] *= w;
      }
#pragma unroll
      for (int k = 0; k < PD; ++k) {
        T v = T(0);
#pragma unroll
        for (int rr = 0; rr < RD; ++rr) {
          const int kk = CR + k * RD + rr;
          v += buf[kk / VEC][kk % VEC] * u[rr];
        }
        o[k] = v;
      }
    }
    for (int off = 1; off < 64; off <<= 1) {
      const int ppt = __shfl_up(pt, off, 64);
      const bool join = lane >= off && ppt == pt;
      if (__ballot(join) == 0ull) break;
      T a[PD];
      for (int k = 0; k < PD; ++k) a[k] = __shfl_up(o[k], off, 64);
      if (join)
        for (int k = 0; k < PD; ++k) o[k] += a[k];
    }
    const int nextPt = __shfl_down(pt, 1, 64);
    const bool tail = active && (lane == 63 || nextPt != pt || j0 == nL - 1);
    if (tail)
      for (int k = 0; k < PD; ++k) atomicAdd(&out[PD * pt + k], o[k]);
  }
}

// Scan-free E^T x variant (env MEGBA_ETX_ATOMIC=1): PD atomicAdds per
// edge instead of the wave segmented scan + tail atomics.  The scan costs
// ~16 dependent shuffle instructions per edge; the atomics serialize on
// the ~degree-5 same-point runs.  Which wins is measured, not assumed.
template <typename T, int CD, int PD, int RD, bool HASINFO>
__global__ void kSpmvEtxPkAtomic(int64_t nL, const int* __restrict__ camOf,
                                 const int* __restrict__ ptOf,
                                 const T* __restrict__ Jpk,
                                 const T* const* __restrict__ jSlots,
                                 const T* __restrict__ info, int lossKind,
                                 T lossD2, const T* __restrict__ xPad,
                                 T* __restrict__ out) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int RW = RD * (RD + 1) / 2;
  constexpr int CR = CD * RD, PR = PD * RD;
  constexpr int NG = (CR + PR + VEC - 1) / VEC;
  const T* rBak = jSlots[2];
  for (int64_t j = blockIdx.x * (int64_t)kBlk + threadIdx.x; j < nL;
       j += (int64_t)gridDim.x * kBlk) {
    const int pt = ptOf[j];
    constexpr int XP = (CD + VEC - 1) / VEC * VEC;
    const TV* xv4 = (const TV*)(xPad + (int64_t)camOf[j] * XP);
    TV xbuf[XP / VEC];
#pragma unroll
    for (int l = 0; l < XP / VEC; ++l) xbuf[l] = xv4[l];
    TV buf[NG];
    const TV* src = (const TV*)Jpk;
#pragma unroll
    for (int g = 0; g < NG; ++g) buf[g] = src[(int64_t)g * nL + j];
    T u[RD];
#pragma unroll
    for (int rr = 0; rr < RD; ++rr) {
      T v = T(0);
#pragma unroll
      for (int i = 0; i < CD; ++i) {
        const int k = i * RD + rr;
        v += buf[k / VEC][k % VEC] * xbuf[i / VEC][i % VEC];
      }
      u[rr] = v;
    }
    if (HASINFO) {
      T wu[RD];
      for (int i = 0; i < RD; ++i) {
        T v = T(0);
        for (int k = 0; k < RD; ++k)
          v += info[RW * j + symIdx<RD>(i, k)] * u[k];
        wu[i] = v;
      }
      for (int i = 0; i < RD; ++i) u[i] = wu[i];
    }
    if (lossKind) {
      T ss = T(0);
      for (int rr = 0; rr < RD; ++rr) {
        const T rv = rBak[(int64_t)rr * nL + j];
        ss += rv * rv;
      }
      const T w = lossWeight(lossKind, lossD2, ss);
      for (int rr = 0; rr < RD; ++rr) u[rr] *= w;
    }
#pragma unroll
    for (int k = 0; k < PD; ++k) {
      T v = T(0);
#pragma unroll
      for (int rr = 0; rr < RD; ++rr) {
        const int kk = CR + k * RD + rr;
        v += buf[kk / VEC][kk % VEC] * u[rr];
      }
      atomicAdd(&out[PD * pt + k], v);
    }
  }
}

// Packed E w over the cam-sorted [wJc(CR), Jp(PR)] groups.
template <typename T, int CD, int PD, int RD>
__global__ __launch_bounds__(64) void kSpmvExPk(
    int nChunks, const int* __restrict__ chCam, const int* __restrict__ chLo,
    const int* __restrict__ chHi, const int* __restrict__ ptOfCam,
    const T* __restrict__ JCamPk, int64_t nL, const T* __restrict__ w,
    T* __restrict__ out) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int CR = CD * RD, PR = PD * RD;
  constexpr int NG = (CR + PR + VEC - 1) / VEC;
  const int chunk = blockIdx.x;
  if (chunk >= nChunks) return;
  const int cam = chCam[chunk];
  T acc[CD];
  for (int i = 0; i < CD; ++i) acc[i] = T(0);
  const int lo = chLo[chunk], hi = chHi[chunk];
  const TV* src = (const TV*)JCamPk;
  // w is stored 4-padded (stride 4 elements, 16B/32B aligned) so the
  // per-edge gather is 1 (fp32) or 2 (fp64) vector loads instead of PD
  // scalar loads — the gather is the TA-request hot spot of this kernel.
  constexpr int WL = (4 * (int)sizeof(T) + 15) / 16;
  for (int j = lo + (int)threadIdx.x; j < hi; j += 64) {
    const TV* wp4 = (const TV*)(w + (int64_t)ptOfCam[j] * 4);
    TV wbuf[WL];
#pragma unroll
    for (int l = 0; l < WL; ++l) wbuf[l] = wp4[l];
    TV buf[NG];
#pragma unroll
    for (int g = 0; g < NG; ++g) buf[g] = src[(int64_t)g * nL + j];
    T u[RD];
#pragma unroll
    for (int rr = 0; rr < RD; ++rr) {
      T v = T(0);
#pragma unroll
      for (int k = 0; k < PD; ++k) {
        const int kk = CR + k * RD + rr;
        v += buf[kk / VEC][kk % VEC] * wbuf[k / VEC][k % VEC];
      }
      u[rr] = v;
    }
#pragma unroll
    for (int i = 0; i < CD; ++i) {
      T v = T(0);
#pragma unroll
      for (int rr = 0; rr < RD; ++rr) {
        const int k = i * RD + rr;
        v += buf[k / VEC][k % VEC] * u[rr];
      }
      acc[i] += v;
    }
  }
  for (int off = 32; off > 0; off >>= 1)
    for (int i = 0; i < CD; ++i) acc[i] += __shfl_down(acc[i], off, 64);
  if (threadIdx.x == 0) {
    T* oc = out + (int64_t)cam * CD;
    for (int i = 0; i < CD; ++i) atomicAdd(&oc[i], acc[i]);
  }
}

// Cam-sorted packed finalize: slab -> [wJc, Jp] vector groups.
template <typename T, int CD, int PD, int RD, bool HASINFO>
__global__ void kFinalizeCamImpPk(int64_t nL, const T* __restrict__ slab,
                                  T* __restrict__ JCamPk) {
  using L = SlabLayout<CD, PD, RD, false, HASINFO>;
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int CR = CD * RD, PR = PD * RD;
  constexpr int NG = (CR + PR + VEC - 1) / VEC;
  constexpr int woff = HASINFO ? L::WJCOFF : L::JCOFF;
  TV* o = (TV*)JCamPk;
  for (int64_t j = blockIdx.x * (int64_t)kBlk + threadIdx.x; j < nL;
       j += (int64_t)gridDim.x * kBlk) {
    const T* row = slab + j * L::SW;
    for (int g = 0; g < NG; ++g) {
      TV v;
      for (int q = 0; q < VEC; ++q) {
        const int k = g * VEC + q;
        v[q] = k < CR ? row[woff + k]
                      : (k < CR + PR ? row[L::JPOFF + (k - CR)] : T(0));
      }
      o[(int64_t)g * nL + j] = v;
    }
  }
}

// ---------------------------------------------------------------------------
// Damping, block inverse
// ---------------------------------------------------------------------------
template <typename T, int D>
__global__ void kDamp(int64_t nElem, const T* __restrict__ H, T* __restrict__ Hd,
                      T f, const unsigned char* __restrict__ fixed) {
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < nElem;
       i += (int64_t)gridDim.x * kBlk) {
    const int within = (int)(i % (D * D));
    const bool diag = within / D == within % D;
    if (fixed && fixed[i / (D * D)]) {
      // fixed vertex: identity block => deltaX = 0 for it (J columns are
      // already zeroed by the forward kernels)
      Hd[i] = diag ? T(1) : T(0);
      continue;
    }
    const T v = H[i];
    Hd[i] = diag ? v * f : v;
  }
}

template <typename T, int D>
__global__ void kInvert(int nBlk, const T* __restrict__ Hd, T* __restrict__ Hinv,
                        int* fail) {
  for (int64_t b = blockIdx.x * (int64_t)kBlk + threadIdx.x; b < nBlk;
       b += (int64_t)gridDim.x * kBlk) {
    if (!spdInvertPacked<T, D>(Hd + b * D * D, Hinv + b * D * D))
      atomicOr(fail, 1);  // rare: retried by kInvertJitter
  }
}

// Rare path: retry semi-definite blocks with a growing relative diagonal
// jitter (matches the CPU oracle); launched only if kInvert reported failure
// so its register/scratch cost stays off the hot path.
template <typename T, int D>
__global__ void kInvertJitter(int nBlk, const T* __restrict__ Hd,
                              T* __restrict__ Hinv, int* fail) {
  for (int64_t b = blockIdx.x * (int64_t)kBlk + threadIdx.x; b < nBlk;
       b += (int64_t)gridDim.x * kBlk) {
    const T* a = Hd + b * D * D;
    T* out = Hinv + b * D * D;
    if (spdInvertPacked<T, D>(a, out)) continue;
    T buf[D * D];
    T mx = T(0);
    for (int i = 0; i < D; ++i) {
      const T ad = (T)fabs((double)a[i * D + i]);
      mx = ad > mx ? ad : mx;
    }
    const T eps = (mx > T(0) ? mx : T(1)) * T(1e-10);
    bool ok = false;
    T jit = eps;
    for (int k = 0; k < 40 && !ok; ++k, jit *= T(10)) {
      for (int i = 0; i < D * D; ++i) buf[i] = a[i];
      for (int i = 0; i < D; ++i) buf[i * D + i] += jit;
      ok = spdInvertPacked<T, D>(buf, out);
    }
    if (!ok) atomicOr(fail + 1, 1);
  }
}

// ---------------------------------------------------------------------------
// Schur SpMV pieces
// ---------------------------------------------------------------------------
// temp[3*pt] = Hpl^T x per local point (fully local, no collective): one
// thread per primary-order edge (coalesced grad-major Hpl or J reads; the
// replicated camera vector x is L2-resident), wave segmented-scan over the
// point runs, atomics only at run tails.  IMP: matrix-free from J
// (reference C23).
template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__global__ void kSpmvEtx(int64_t nL, const int* __restrict__ camOf,
                         const int* __restrict__ ptOf,
                         const T* __restrict__ Hpl,
                         const T* const* __restrict__ jSlots,
                         const T* __restrict__ info, int lossKind, T lossD2,
                         const T* __restrict__ x, T* __restrict__ out) {
  // IMP reads the accepted J set through the device pointer slots so a
  // captured graph stays valid across accept-time buffer flips.
  constexpr int RW = RD * (RD + 1) / 2;
  const T* Jc = IMP ? jSlots[0] : nullptr;
  const T* Jp = IMP ? jSlots[1] : nullptr;
  const T* rBak = IMP ? jSlots[2] : nullptr;
  const int lane = threadIdx.x & 63;
  const int64_t nWork = ((nL + kBlk - 1) / kBlk) * (int64_t)kBlk;
  for (int64_t j0 = blockIdx.x * (int64_t)kBlk + threadIdx.x; j0 < nWork;
       j0 += (int64_t)gridDim.x * kBlk) {
    const bool active = j0 < nL;
    const int64_t j = active ? j0 : nL - 1;
    const int pt = ptOf[j];
    T o[PD];
    for (int k = 0; k < PD; ++k) o[k] = T(0);
    if (active) {
      const T* xc = x + (int64_t)camOf[j] * CD;
      if (IMP) {
        T u[RD];
        for (int rr = 0; rr < RD; ++rr) {
          T v = T(0);
          for (int i = 0; i < CD; ++i)
            v += Jc[((int64_t)(i * RD + rr)) * nL + j] * xc[i];
          u[rr] = v;
        }
        if (HASINFO) {
          T wu[RD];
          for (int i = 0; i < RD; ++i) {
            T v = T(0);
            for (int k = 0; k < RD; ++k)
              v += info[RW * j + symIdx<RD>(i, k)] * u[k];
            wu[i] = v;
          }
          for (int i = 0; i < RD; ++i) u[i] = wu[i];
        }
        if (lossKind) {
          T ss = T(0);
          for (int rr = 0; rr < RD; ++rr) {
            const T rv = rBak[(int64_t)rr * nL + j];
            ss += rv * rv;
          }
          const T w = lossWeight(lossKind, lossD2, ss);
          for (int rr = 0; rr < RD; ++rr) u[rr] *= w;
        }
        for (int k = 0; k < PD; ++k) {
          T v = T(0);
          for (int rr = 0; rr < RD; ++rr)
            v += Jp[((int64_t)(k * RD + rr)) * nL + j] * u[rr];
          o[k] = v;
        }
      } else {
        using TV = typename PackVec<T>::type;
        constexpr int VEC = PackVec<T>::VEC;
        constexpr int NGH = (CD * PD + VEC - 1) / VEC;
        TV buf[NGH];
        const TV* src = (const TV*)Hpl;
#pragma unroll
        for (int g = 0; g < NGH; ++g) buf[g] = src[(int64_t)g * nL + j];
#pragma unroll
        for (int i = 0; i < CD; ++i) {
          const T xi = xc[i];
#pragma unroll
          for (int k = 0; k < PD; ++k) {
            const int kk = i * PD + k;
            o[k] += buf[kk / VEC][kk % VEC] * xi;
          }
        }
      }
    }
    for (int off = 1; off < 64; off <<= 1) {
      const int ppt = __shfl_up(pt, off, 64);
      const bool join = lane >= off && ppt == pt;
      // Point runs are short (degree ~5): once no lane continues a segment
      // at distance `off`, no lane can at any larger distance either — skip
      // the remaining dependent shuffle rounds (the scan chain is this
      // kernel's issue-stall bound, 70% SQ_WAIT_INST_ANY).
      if (__ballot(join) == 0ull) break;
      T a[PD];
      for (int k = 0; k < PD; ++k) a[k] = __shfl_up(o[k], off, 64);
      if (join)
        for (int k = 0; k < PD; ++k) o[k] += a[k];
    }
    const int nextPt = __shfl_down(pt, 1, 64);
    const bool tail = active && (lane == 63 || nextPt != pt || j0 == nL - 1);
    if (tail)
      for (int k = 0; k < PD; ++k) atomicAdd(&out[PD * pt + k], o[k]);
  }
}

// out[9*cam] += E w partials: one wave per <=256-row chunk of one camera's
// cam-sorted rows; per-lane partials, wave-wide shuffle reduce, one
// atomicAdd set per chunk.  Caller allreduces 9*ncam (the ONLY per-PCG-
// iteration collective).
template <typename T, int CD, int PD>
__global__ __launch_bounds__(64) void kSpmvEx(
    int nChunks, const int* __restrict__ chCam, const int* __restrict__ chLo,
    const int* __restrict__ chHi, const int* __restrict__ ptOfCam,
    const T* __restrict__ HplCam, int64_t nL, const T* __restrict__ w,
    T* __restrict__ out) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int NGH = (CD * PD + VEC - 1) / VEC;
  constexpr int WL = (4 * (int)sizeof(T) + 15) / 16;
  const int chunk = blockIdx.x;
  if (chunk >= nChunks) return;
  const int cam = chCam[chunk];
  T acc[CD];
  for (int i = 0; i < CD; ++i) acc[i] = T(0);
  const int lo = chLo[chunk], hi = chHi[chunk];
  const TV* src = (const TV*)HplCam;
  for (int j = lo + (int)threadIdx.x; j < hi; j += 64) {
    const TV* wp4 = (const TV*)(w + (int64_t)ptOfCam[j] * 4);
    TV wbuf[WL];
#pragma unroll
    for (int l = 0; l < WL; ++l) wbuf[l] = wp4[l];
    TV buf[NGH];
#pragma unroll
    for (int g = 0; g < NGH; ++g) buf[g] = src[(int64_t)g * nL + j];
#pragma unroll
    for (int i = 0; i < CD; ++i) {
      T v = T(0);
#pragma unroll
      for (int k = 0; k < PD; ++k) {
        const int kk = i * PD + k;
        v += buf[kk / VEC][kk % VEC] * wbuf[k / VEC][k % VEC];
      }
      acc[i] += v;
    }
  }
  for (int off = 32; off > 0; off >>= 1)
    for (int i = 0; i < CD; ++i) acc[i] += __shfl_down(acc[i], off, 64);
  if (threadIdx.x == 0) {
    T* oc = out + (int64_t)cam * CD;
    for (int i = 0; i < CD; ++i) atomicAdd(&oc[i], acc[i]);
  }
}

// ---------------------------------------------------------------------------
// Fused Schur apply:  out[CD*ncam] += E Cinv E^T x  in ONE pass.
// ---------------------------------------------------------------------------
// The separate-pass pipeline (kSpmvEtx -> applyCinv -> kSpmvEx) reads the
// per-edge J/Hpl data twice per PCG iteration and gathers the 3*npt w
// vector by point id from the cam-sorted pass — on big-fp32 problems that
// gather fetches a full 64 B line for 12 used bytes and was measured at
// ~2x the byte floor (profiles/r01_final13682_implicit_fp32.md).  Here the
// edges are processed in primary (pt,cam)-sorted order in WINDOWS of <=64
// edges aligned to point-run boundaries (host-built table), one wave per
// window:
//   1. each lane evaluates its edge's t_e = Jp^T W (Jc x)  (x is the
//      replicated CD*ncam vector, L2-resident),
//   2. a wave segmented-scan sums t_e over each point run; the run tail
//      applies the PD x PD Cinv block: w_p = Cinv t_p,
//   3. w_p is broadcast back to the run's lanes (tail-lane shuffle) and
//      each lane scatters its edge's E-contribution Jc^T W (Jp w_p) with
//      CD atomicAdds into the small L2-resident camera vector.
// J (or Hpl) is read ONCE, w never exists in HBM, and the scan/broadcast
// replace both the temp round-trip and the gather.  Point runs longer
// than 64 edges (high-degree landmarks) fall back to the classic path:
// their windows only accumulate t partials into `temp` (phase 1 flag);
// kFusedLongW then applies Cinv for those points and kFusedLongScatter
// finishes their E-contributions.
template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__device__ inline void fusedLoadEdge(int64_t j, int64_t nL,
                                     const int* __restrict__ camOf,
                                     const T* __restrict__ Hpl,
                                     const T* __restrict__ Jc,
                                     const T* __restrict__ Jp,
                                     const T* __restrict__ info,
                                     const T* __restrict__ rBak, int lossKind,
                                     T lossD2, const T* __restrict__ xPad,
                                     T (&jcv)[RD][CD], T (&jpv)[RD][PD],
                                     T (&hplv)[CD][PD], T (&t)[PD]) {
  constexpr int RW = RD * (RD + 1) / 2;
  constexpr int XPV = PackVec<T>::VEC;
  constexpr int XP = (CD + XPV - 1) / XPV * XPV;
  const T* xc = xPad + (int64_t)camOf[j] * XP;
  if (IMP) {
    for (int i = 0; i < CD; ++i)
      for (int rr = 0; rr < RD; ++rr)
        jcv[rr][i] = Jc[((int64_t)(i * RD + rr)) * nL + j];
    for (int k = 0; k < PD; ++k)
      for (int rr = 0; rr < RD; ++rr)
        jpv[rr][k] = Jp[((int64_t)(k * RD + rr)) * nL + j];
    T u[RD];
    for (int rr = 0; rr < RD; ++rr) {
      T v = T(0);
      for (int i = 0; i < CD; ++i) v += jcv[rr][i] * xc[i];
      u[rr] = v;
    }
    if (HASINFO) {
      T wu[RD];
      for (int i = 0; i < RD; ++i) {
        T v = T(0);
        for (int k = 0; k < RD; ++k)
          v += info[RW * j + symIdx<RD>(i, k)] * u[k];
        wu[i] = v;
      }
      for (int i = 0; i < RD; ++i) u[i] = wu[i];
    }
    if (lossKind) {
      T ss = T(0);
      for (int rr = 0; rr < RD; ++rr) {
        const T rv = rBak[(int64_t)rr * nL + j];
        ss += rv * rv;
      }
      const T w = lossWeight(lossKind, lossD2, ss);
      for (int rr = 0; rr < RD; ++rr) u[rr] *= w;
    }
    for (int k = 0; k < PD; ++k) {
      T v = T(0);
      for (int rr = 0; rr < RD; ++rr) v += jpv[rr][k] * u[rr];
      t[k] = v;
    }
  } else {
    using TV = typename PackVec<T>::type;
    constexpr int VEC = PackVec<T>::VEC;
    constexpr int NGH = (CD * PD + VEC - 1) / VEC;
    const TV* src = (const TV*)Hpl;
    TV buf[NGH];
    for (int g = 0; g < NGH; ++g) buf[g] = src[(int64_t)g * nL + j];
    for (int i = 0; i < CD; ++i)
      for (int k = 0; k < PD; ++k) {
        const int kk = i * PD + k;
        hplv[i][k] = buf[kk / VEC][kk % VEC];
      }
    for (int k = 0; k < PD; ++k) t[k] = T(0);
    for (int i = 0; i < CD; ++i) {
      const T xi = xc[i];
      for (int k = 0; k < PD; ++k) t[k] += hplv[i][k] * xi;
    }
  }
}

// Per-edge E-contribution out[cam] += M w  (M = Jc^T W Jp or Hpl).
template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__device__ inline void fusedScatter(int64_t j, int64_t nL,
                                    const int* __restrict__ camOf,
                                    const T* __restrict__ info,
                                    const T* __restrict__ rBak, int lossKind,
                                    T lossD2, const T (&jcv)[RD][CD],
                                    const T (&jpv)[RD][PD],
                                    const T (&hplv)[CD][PD], const T (&wv)[PD],
                                    T* __restrict__ out) {
  constexpr int RW = RD * (RD + 1) / 2;
  T* oc = out + (int64_t)camOf[j] * CD;
  if (IMP) {
    T u[RD];
    for (int rr = 0; rr < RD; ++rr) {
      T v = T(0);
      for (int k = 0; k < PD; ++k) v += jpv[rr][k] * wv[k];
      u[rr] = v;
    }
    if (HASINFO) {
      T wu[RD];
      for (int i = 0; i < RD; ++i) {
        T v = T(0);
        for (int k = 0; k < RD; ++k)
          v += info[RW * j + symIdx<RD>(i, k)] * u[k];
        wu[i] = v;
      }
      for (int i = 0; i < RD; ++i) u[i] = wu[i];
    }
    if (lossKind) {
      T ss = T(0);
      for (int rr = 0; rr < RD; ++rr) {
        const T rv = rBak[(int64_t)rr * nL + j];
        ss += rv * rv;
      }
      const T w = lossWeight(lossKind, lossD2, ss);
      for (int rr = 0; rr < RD; ++rr) u[rr] *= w;
    }
    for (int i = 0; i < CD; ++i) {
      T v = T(0);
      for (int rr = 0; rr < RD; ++rr) v += jcv[rr][i] * u[rr];
      atomicAdd(&oc[i], v);
    }
  } else {
    for (int i = 0; i < CD; ++i) {
      T v = T(0);
      for (int k = 0; k < PD; ++k) v += hplv[i][k] * wv[k];
      atomicAdd(&oc[i], v);
    }
  }
}

template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__global__ __launch_bounds__(256) void kSchurFused(
    int nWin, const int64_t* __restrict__ winLo,
    const int64_t* __restrict__ winHi, const unsigned char* __restrict__ winFlag,
    int64_t nL, const int* __restrict__ camOf, const int* __restrict__ ptOf,
    const T* __restrict__ Hpl, const T* const* __restrict__ jSlots,
    const T* __restrict__ info, int lossKind, T lossD2,
    const T* __restrict__ xPad, const T* __restrict__ HllInv,
    T* __restrict__ temp, T* __restrict__ out) {
  constexpr int PP = PD * PD;
  const T* Jc = IMP ? jSlots[0] : nullptr;
  const T* Jp = IMP ? jSlots[1] : nullptr;
  const T* rBak = IMP ? jSlots[2] : nullptr;
  const int lane = threadIdx.x & 63;
  for (int w = blockIdx.x * 4 + ((int)threadIdx.x >> 6); w < nWin;
       w += (int)gridDim.x * 4) {
    const int64_t lo = winLo[w], hi = winHi[w];
    const bool active = lo + lane < hi;
    const int64_t j = active ? lo + lane : hi - 1;
    const int pt = ptOf[j];
    T jcv[RD][CD], jpv[RD][PD], hplv[CD][PD], t[PD];
    for (int k = 0; k < PD; ++k) t[k] = T(0);
    if (active)
      fusedLoadEdge<T, CD, PD, RD, IMP, HASINFO>(
          j, nL, camOf, Hpl, Jc, Jp, info, rBak, lossKind, lossD2, xPad,
          jcv, jpv, hplv, t);
    // segmented inclusive scan over the window's point runs
    for (int off = 1; off < 64; off <<= 1) {
      const int ppt = __shfl_up(pt, off, 64);
      const bool join = lane >= off && ppt == pt;
      if (__ballot(join) == 0ull) break;
      T a[PD];
      for (int k = 0; k < PD; ++k) a[k] = __shfl_up(t[k], off, 64);
      if (join)
        for (int k = 0; k < PD; ++k) t[k] += a[k];
    }
    const int nextPt = __shfl_down(pt, 1, 64);
    const bool tail = active && (lo + lane == hi - 1 || nextPt != pt);
    if (winFlag[w]) {
      // segment of a >64-edge run: partials only (finished by the long-run
      // kernels)
      if (tail)
        for (int k = 0; k < PD; ++k) atomicAdd(&temp[PD * pt + k], t[k]);
      continue;
    }
    T wv[PD];
    for (int k = 0; k < PD; ++k) wv[k] = T(0);
    if (tail) {
      const T* inv = HllInv + (int64_t)pt * PP;
      for (int i = 0; i < PD; ++i) {
        T v = T(0);
        for (int k = 0; k < PD; ++k) v += inv[i * PD + k] * t[k];
        wv[i] = v;
      }
    }
    // broadcast w_p from the run tail back to every lane of the run:
    // tail lanes form a mask; this lane's tail is the first tail at or
    // after it (points are sorted within the window).
    const unsigned long long tails = __ballot(tail);
    const unsigned long long from = tails >> lane;
    const int tl = from ? lane + __ffsll((long long)from) - 1 : lane;
    for (int k = 0; k < PD; ++k) wv[k] = __shfl(wv[k], tl, 64);
    if (active)
      fusedScatter<T, CD, PD, RD, IMP, HASINFO>(
          j, nL, camOf, info, rBak, lossKind, lossD2, jcv, jpv, hplv, wv,
          out);
  }
}

// Fused E^T x + Cinv over run-aligned windows (the winning half of the
// kSchurFused experiment): each window's runs are complete, so the run
// TAIL applies the PD x PD Cinv block to the scanned t_p and stores
// w_p = Cinv E^T x straight into the 4-padded w vector — no temp
// round-trip, no zeroing, no atomics, and the separate Cinv pass
// disappears.  Long (>64-edge) runs fall back to temp partials finished
// by kFusedLongW.  E w then proceeds from wPad as usual (the gather-side
// alternative, a fused scatter, measured 5x worse — Experiment 1).
template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__global__ __launch_bounds__(256) void kEtxCinvFused(
    int nWin, const int64_t* __restrict__ winLo,
    const int64_t* __restrict__ winHi, const unsigned char* __restrict__ winFlag,
    int64_t nL, const int* __restrict__ camOf, const int* __restrict__ ptOf,
    const T* __restrict__ Hpl, const T* const* __restrict__ jSlots,
    const T* __restrict__ info, int lossKind, T lossD2,
    const T* __restrict__ xPad, const T* __restrict__ HllInv,
    T* __restrict__ temp, T* __restrict__ wPad) {
  constexpr int PP = PD * PD;
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int WL = (4 + VEC - 1) / VEC;
  const T* Jc = IMP ? jSlots[0] : nullptr;
  const T* Jp = IMP ? jSlots[1] : nullptr;
  const T* rBak = IMP ? jSlots[2] : nullptr;
  const int lane = threadIdx.x & 63;
  for (int w = blockIdx.x * 4 + ((int)threadIdx.x >> 6); w < nWin;
       w += (int)gridDim.x * 4) {
    const int64_t lo = winLo[w], hi = winHi[w];
    const bool active = lo + lane < hi;
    const int64_t j = active ? lo + lane : hi - 1;
    const int pt = ptOf[j];
    T jcv[RD][CD], jpv[RD][PD], hplv[CD][PD], t[PD];
    for (int k = 0; k < PD; ++k) t[k] = T(0);
    if (active)
      fusedLoadEdge<T, CD, PD, RD, IMP, HASINFO>(
          j, nL, camOf, Hpl, Jc, Jp, info, rBak, lossKind, lossD2, xPad,
          jcv, jpv, hplv, t);
    for (int off = 1; off < 64; off <<= 1) {
      const int ppt = __shfl_up(pt, off, 64);
      const bool join = lane >= off && ppt == pt;
      if (__ballot(join) == 0ull) break;
      T a[PD];
      for (int k = 0; k < PD; ++k) a[k] = __shfl_up(t[k], off, 64);
      if (join)
        for (int k = 0; k < PD; ++k) t[k] += a[k];
    }
    const int nextPt = __shfl_down(pt, 1, 64);
    const bool tail = active && (lo + lane == hi - 1 || nextPt != pt);
    if (winFlag[w]) {
      if (tail)
        for (int k = 0; k < PD; ++k) atomicAdd(&temp[PD * pt + k], t[k]);
      continue;
    }
    if (tail) {
      const T* inv = HllInv + (int64_t)pt * PP;
      T out[WL * VEC];
      for (int k = 0; k < WL * VEC; ++k) out[k] = T(0);
      for (int i = 0; i < PD; ++i) {
        T v = T(0);
        for (int k = 0; k < PD; ++k) v += inv[i * PD + k] * t[k];
        out[i] = v;
      }
      TV* o = (TV*)(wPad + (int64_t)pt * 4);
      for (int l = 0; l < WL; ++l) {
        TV v;
        for (int q = 0; q < VEC; ++q) v[q] = out[l * VEC + q];
        o[l] = v;
      }
    }
  }
}

// Long-run phase 2: w_p = Cinv temp_p for the listed high-degree points,
// stored into the 4-padded w vector.
template <typename T, int PD>
__global__ void kFusedLongW(int nPts, const int* __restrict__ longPts,
                            const T* __restrict__ HllInv,
                            const T* __restrict__ temp, T* __restrict__ w,
                            int wStride) {
  constexpr int PP = PD * PD;
  for (int idx = blockIdx.x * (int)blockDim.x + (int)threadIdx.x; idx < nPts;
       idx += (int)gridDim.x * (int)blockDim.x) {
    const int pt = longPts[idx];
    const T* inv = HllInv + (int64_t)pt * PP;
    for (int i = 0; i < PD; ++i) {
      T v = T(0);
      for (int k = 0; k < PD; ++k) v += inv[i * PD + k] * temp[PD * pt + k];
      w[(int64_t)wStride * pt + i] = v;
    }
  }
}

// Long-run phase 3: finish the flagged windows' E-contributions with w
// read from the global vector.
template <typename T, int CD, int PD, int RD, bool IMP, bool HASINFO>
__global__ __launch_bounds__(256) void kFusedLongScatter(
    int nFlag, const int* __restrict__ flagWins,
    const int64_t* __restrict__ winLo, const int64_t* __restrict__ winHi,
    int64_t nL, const int* __restrict__ camOf, const int* __restrict__ ptOf,
    const T* __restrict__ Hpl, const T* const* __restrict__ jSlots,
    const T* __restrict__ info, int lossKind, T lossD2,
    const T* __restrict__ xPad, const T* __restrict__ w,
    T* __restrict__ out) {
  const T* Jc = IMP ? jSlots[0] : nullptr;
  const T* Jp = IMP ? jSlots[1] : nullptr;
  const T* rBak = IMP ? jSlots[2] : nullptr;
  const int lane = threadIdx.x & 63;
  for (int f = blockIdx.x * 4 + ((int)threadIdx.x >> 6); f < nFlag;
       f += (int)gridDim.x * 4) {
    const int wi = flagWins[f];
    const int64_t lo = winLo[wi], hi = winHi[wi];
    if (lo + lane >= hi) continue;
    const int64_t j = lo + lane;
    const int pt = ptOf[j];
    T jcv[RD][CD], jpv[RD][PD], hplv[CD][PD], t[PD];
    fusedLoadEdge<T, CD, PD, RD, IMP, HASINFO>(
        j, nL, camOf, Hpl, Jc, Jp, info, rBak, lossKind, lossD2, xPad, jcv,
        jpv, hplv, t);
    T wv[PD];
    for (int k = 0; k < PD; ++k) wv[k] = w[PD * pt + k];
    fusedScatter<T, CD, PD, RD, IMP, HASINFO>(
        j, nL, camOf, info, rBak, lossKind, lossD2, jcv, jpv, hplv, wv,
        out);
  }
}

// Fused preconditioner apply + rho partial: z = Binv r (thread per row) and
// per-block partials of r.z in one pass (saves two launches per PCG iter).
template <typename T, int CD>
__global__ void kPrecondRho(int nBlk, const T* __restrict__ Binv,
                            const T* __restrict__ r, T* __restrict__ z,
                            double* part) {
  __shared__ double sm[kBlk];
  double local = 0.0;
  for (int64_t idx = blockIdx.x * (int64_t)kBlk + threadIdx.x;
       idx < (int64_t)nBlk * CD; idx += (int64_t)gridDim.x * kBlk) {
    const int64_t b = idx / CD;
    const int rr = (int)(idx % CD);
    const T* row = Binv + b * CD * CD + (int64_t)rr * CD;
    const T* xb = r + b * CD;
    T sv = T(0);
    for (int j = 0; j < CD; ++j) sv += row[j] * xb[j];
    z[idx] = sv;
    local += (double)sv * (double)r[idx];
  }
  sm[threadIdx.x] = local;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) sm[threadIdx.x] += sm[threadIdx.x + st];
    __syncthreads();
  }
  if (threadIdx.x == 0) part[blockIdx.x] = sm[0];
}

// B-apply (y = A x - y, the Schur S-apply tail) fused with the per-block
// p^T q dot partials: saves the separate full pass over p and q per PCG
// iteration (the reference used a standalone cublasDot, its :368-385).
template <typename T, int CD>
__global__ void kBApplyDot(int nBlk, const T* __restrict__ A,
                           const T* __restrict__ x, T* __restrict__ y,
                           double* part) {
  __shared__ double sm[kBlk];
  double local = 0.0;
  for (int64_t idx = blockIdx.x * (int64_t)kBlk + threadIdx.x;
       idx < (int64_t)nBlk * CD; idx += (int64_t)gridDim.x * kBlk) {
    const int64_t b = idx / CD;
    const int rrow = (int)(idx % CD);
    const T* row = A + b * CD * CD + (int64_t)rrow * CD;
    const T* xb = x + b * CD;
    T s = T(0);
    for (int j = 0; j < CD; ++j) s += row[j] * xb[j];
    const T q = s - y[idx];
    y[idx] = q;
    local += (double)x[idx] * (double)q;
  }
  sm[threadIdx.x] = local;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) sm[threadIdx.x] += sm[threadIdx.x + st];
    __syncthreads();
  }
  if (threadIdx.x == 0) part[blockIdx.x] = sm[0];
}

// Block-diagonal matvec, one thread per output row.
// MODE 0: y = A x;  MODE 1: y = A x - y  (the reference's rw=1,dw=-1 gemv).
template <typename T, int D, int MODE>
__global__ void kBlockDiagMatVec(int nBlk, const T* __restrict__ A,
                                 const T* __restrict__ x, T* __restrict__ y) {
  for (int64_t idx = blockIdx.x * (int64_t)kBlk + threadIdx.x;
       idx < (int64_t)nBlk * D; idx += (int64_t)gridDim.x * kBlk) {
    const int64_t b = idx / D;
    const int rrow = (int)(idx % D);
    const T* row = A + b * D * D + (int64_t)rrow * D;
    const T* xb = x + b * D;
    T s = T(0);
    for (int j = 0; j < D; ++j) s += row[j] * xb[j];
    y[idx] = (MODE == 0) ? s : s - y[idx];
  }
}

// Cinv apply writing the 4-padded w layout read by kSpmvExPk: one thread
// per point, whole padded slot written as 16/32-byte vector stores.
template <typename T, int PD>
__global__ void kCinvPad(int nBlk, const T* __restrict__ A,
                         const T* __restrict__ x, T* __restrict__ yPad) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int WL = (4 + VEC - 1) / VEC;
  for (int64_t b = blockIdx.x * (int64_t)kBlk + threadIdx.x; b < nBlk;
       b += (int64_t)gridDim.x * kBlk) {
    const T* inv = A + b * PD * PD;
    const T* xb = x + b * PD;
    T out[WL * VEC];
    for (int k = 0; k < WL * VEC; ++k) out[k] = T(0);
    for (int i = 0; i < PD; ++i) {
      T sv = T(0);
      for (int j = 0; j < PD; ++j) sv += inv[i * PD + j] * xb[j];
      out[i] = sv;
    }
    TV* o = (TV*)(yPad + b * 4);
    for (int l = 0; l < WL; ++l) {
      TV v;
      for (int q = 0; q < VEC; ++q) v[q] = out[l * VEC + q];
      o[l] = v;
    }
  }
}

// Pad the CD-stride camera vector to a 16B-aligned XP stride (one vector
// store per camera) so the E^T x per-edge gather is XP/VEC vector loads
// instead of CD divergent scalar loads.
template <typename T, int CD>
__global__ void kPadX(int ncam, const T* __restrict__ x,
                      T* __restrict__ xPad) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int XP = (CD + VEC - 1) / VEC * VEC;
  for (int64_t c = blockIdx.x * (int64_t)kBlk + threadIdx.x; c < ncam;
       c += (int64_t)gridDim.x * kBlk) {
    const T* xc = x + c * CD;
    TV* o = (TV*)(xPad + c * XP);
    for (int l = 0; l < XP / VEC; ++l) {
      TV v;
      for (int q = 0; q < VEC; ++q) {
        const int k = l * VEC + q;
        v[q] = k < CD ? xc[k] : T(0);
      }
      o[l] = v;
    }
  }
}

// ---------------------------------------------------------------------------
// Small vector kernels
// ---------------------------------------------------------------------------
// Device-side pointer slots for the implicit J buffers: the double-buffered
// accepted set flips on every LM accept, but the captured PCG graph freezes
// kernel arguments — so the kernels dereference this slot array instead and
// acceptForward rewrites it (kernel args carry the values; no host-buffer
// lifetime to manage).  Slots: [0]=Jc, [1]=Jp, [2]=r (all accepted/bak).
template <typename T>
__global__ void kSetPtrSlots(const T** slots, const T* a, const T* b,
                             const T* c) {
  slots[0] = a;
  slots[1] = b;
  slots[2] = c;
}

__global__ void kSetScalar(double* out, double v) { *out = v; }
// (The standalone kRedFinalRhoBeta / kRedFinalAlpha reduction finals were
// folded into kXpbySBeta / kUpdateXRAlpha below in r2.)

// Fused beta + p-update: every block redundantly reduces the (tiny)
// rho partial array to beta = rho/rhoPrev, then updates its slice of
// p = z + beta p.  Replaces the standalone kRedFinalRhoBeta launch
// (block 0 still publishes rho for the host refuse/tol readback).
template <typename T>
__global__ void kXpbySBeta(int64_t n, const T* __restrict__ z,
                           const double* __restrict__ part, int nb,
                           const double* __restrict__ rhoPrev,
                           double* __restrict__ rhoOut, T* __restrict__ p) {
  __shared__ double sm[kBlk];
  double v = 0.0;
  for (int i = threadIdx.x; i < nb; i += kBlk) v += part[i];
  sm[threadIdx.x] = v;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) sm[threadIdx.x] += sm[threadIdx.x + st];
    __syncthreads();
  }
  const double rho = sm[0];
  if (blockIdx.x == 0 && threadIdx.x == 0) *rhoOut = rho;
  const double rp = *rhoPrev;
  const T beta = (T)(rp != 0.0 ? rho / rp : 0.0);
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk)
    p[i] = z[i] + beta * p[i];
}

// Fused alpha + x/r update (incl. the two-deep backup rotation): every
// block reduces BOTH partial arrays (rho from partR, p^T q from partQ),
// forms alpha = rho/pq, and updates its slice; block 0 publishes
// rhoPrev = rho for the next iteration's beta.
template <typename T>
__global__ void kUpdateXRAlpha(int64_t n, const double* __restrict__ partR,
                               int nbR, const double* __restrict__ partQ,
                               int nbQ, double* __restrict__ rhoPrevOut,
                               const T* __restrict__ p,
                               const T* __restrict__ q, T* __restrict__ x,
                               T* __restrict__ xBak,
                               T* __restrict__ xBakPrev, T* __restrict__ r) {
  __shared__ double smR[kBlk];
  __shared__ double smQ[kBlk];
  double vR = 0.0, vQ = 0.0;
  for (int i = threadIdx.x; i < nbR; i += kBlk) vR += partR[i];
  for (int i = threadIdx.x; i < nbQ; i += kBlk) vQ += partQ[i];
  smR[threadIdx.x] = vR;
  smQ[threadIdx.x] = vQ;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) {
      smR[threadIdx.x] += smR[threadIdx.x + st];
      smQ[threadIdx.x] += smQ[threadIdx.x + st];
    }
    __syncthreads();
  }
  const double rho = smR[0];
  const double pq = smQ[0];
  if (blockIdx.x == 0 && threadIdx.x == 0) *rhoPrevOut = rho;
  const T a = (T)(pq != 0.0 ? rho / pq : 0.0);
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk) {
    xBakPrev[i] = xBak[i];
    const T xv = x[i];
    xBak[i] = xv;
    x[i] = xv + a * p[i];
    r[i] -= a * q[i];
  }
}

// r = v - q
template <typename T>
__global__ void kSub(int64_t n, const T* __restrict__ v, const T* __restrict__ q,
                     T* __restrict__ r) {
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk)
    r[i] = v[i] - q[i];
}
// v = gc * s - v   (s = 1/worldSize pre-compensation, reference :478)
template <typename T>
__global__ void kVMake(int64_t n, const T* __restrict__ gc, T s, T* __restrict__ v) {
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk)
    v[i] = gc[i] * s - v[i];
}
template <typename T>
__global__ void kAddAssign(int64_t n, const T* __restrict__ x, T* __restrict__ y) {
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk)
    y[i] += x[i];
}
template <typename T>
__global__ void kZeroRange(T* p, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)kBlk + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * kBlk)
    p[i] = T(0);
}
// deltaX_p = HllInv * (g_p - temp)  (local point shard)
template <typename T, int PD>
__global__ void kBackSub(int npt, const T* __restrict__ HllInv,
                         const T* __restrict__ gp, const T* __restrict__ temp,
                         T* __restrict__ dxp) {
  for (int64_t p = blockIdx.x * (int64_t)kBlk + threadIdx.x; p < npt;
       p += (int64_t)gridDim.x * kBlk) {
    const T* inv = HllInv + p * PD * PD;
    T rhs[PD];
    for (int i = 0; i < PD; ++i) rhs[i] = gp[PD * p + i] - temp[PD * p + i];
    for (int i = 0; i < PD; ++i) {
      T v = T(0);
      for (int j = 0; j < PD; ++j) v += inv[i * PD + j] * rhs[j];
      dxp[PD * p + i] = v;
    }
  }
}

// ---------------------------------------------------------------------------
// rho denominator:  sum over edges of (J dx + r)^2   (backup J/r)
// ---------------------------------------------------------------------------
template <typename T, int CD, int PD, int RD>
__global__ void kRhoDenom(int64_t nL, const int* __restrict__ camOf,
                          const int* __restrict__ ptOf, const T* __restrict__ r,
                          const T* __restrict__ Jc, const T* __restrict__ Jp,
                          const T* __restrict__ dxc, const T* __restrict__ dxp,
                          double* acc, int lossKind, T lossD2) {
  __shared__ double sm[kBlk];
  double local = 0.0;
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    const T* dc = dxc + (int64_t)camOf[e] * CD;
    const T* dp = dxp + (int64_t)ptOf[e] * PD;
    T ss = T(0);
    for (int row = 0; row < RD; ++row) {
      T s = r[(int64_t)row * nL + e];
      for (int k = 0; k < CD; ++k)
        s += Jc[((int64_t)(k * RD + row)) * nL + e] * dc[k];
      for (int k = 0; k < PD; ++k)
        s += Jp[((int64_t)(k * RD + row)) * nL + e] * dp[k];
      ss += s * s;
    }
    local += (double)lossRho(lossKind, lossD2, ss);
  }
  sm[threadIdx.x] = local;
  __syncthreads();
  for (int s = kBlk / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) sm[threadIdx.x] += sm[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(acc, sm[0]);
}

// Packed-J rho denominator (implicit mode): the accepted [Jc, Jp] vector
// groups and the XP-padded deltaX camera vector replace 26 scalar loads
// per edge (same math as kRhoDenom).
template <typename T, int CD, int PD, int RD>
__global__ void kRhoDenomPk(int64_t nL, const int* __restrict__ camOf,
                            const int* __restrict__ ptOf,
                            const T* __restrict__ r,
                            const T* __restrict__ Jpk,
                            const T* __restrict__ dxcPad,
                            const T* __restrict__ dxp, double* acc,
                            int lossKind, T lossD2) {
  using TV = typename PackVec<T>::type;
  constexpr int VEC = PackVec<T>::VEC;
  constexpr int CR = CD * RD, PR = PD * RD;
  constexpr int NG = (CR + PR + VEC - 1) / VEC;
  constexpr int XP = (CD + VEC - 1) / VEC * VEC;
  __shared__ double sm[kBlk];
  double local = 0.0;
  const TV* src = (const TV*)Jpk;
  for (int64_t e = blockIdx.x * (int64_t)kBlk + threadIdx.x; e < nL;
       e += (int64_t)gridDim.x * kBlk) {
    const TV* xv4 = (const TV*)(dxcPad + (int64_t)camOf[e] * XP);
    TV xbuf[XP / VEC];
#pragma unroll
    for (int l = 0; l < XP / VEC; ++l) xbuf[l] = xv4[l];
    TV buf[NG];
#pragma unroll
    for (int g = 0; g < NG; ++g) buf[g] = src[(int64_t)g * nL + e];
    const T* dp = dxp + (int64_t)ptOf[e] * PD;
    T ss = T(0);
#pragma unroll
    for (int row = 0; row < RD; ++row) {
      T sv = r[(int64_t)row * nL + e];
#pragma unroll
      for (int k = 0; k < CD; ++k) {
        const int kk = k * RD + row;
        sv += buf[kk / VEC][kk % VEC] * xbuf[k / VEC][k % VEC];
      }
#pragma unroll
      for (int k = 0; k < PD; ++k) {
        const int kk = CR + k * RD + row;
        sv += buf[kk / VEC][kk % VEC] * dp[k];
      }
      ss += sv * sv;
    }
    local += (double)lossRho(lossKind, lossD2, ss);
  }
  sm[threadIdx.x] = local;
  __syncthreads();
  for (int st = kBlk / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) sm[threadIdx.x] += sm[threadIdx.x + st];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(acc, sm[0]);
}

}  // namespace

// ===========================================================================
// Host engine
// ===========================================================================
template <typename T, int CD, int PD, int RD>
class GpuEngine final : public Engine<T> {
  static constexpr int GW = CD + PD;            // gradient width
  static constexpr int CC = CD * CD;            // Hpp block
  static constexpr int PP = PD * PD;            // Hll block
  static constexpr int CP = CD * PD;            // Hpl block
  static constexpr int CR = CD * RD;            // Jc values per edge
  static constexpr int PR = PD * RD;            // Jp values per edge
  static constexpr int RW = RD * (RD + 1) / 2;  // packed info entries

 public:
  GpuEngine(const BAProblemHost& prob, const ProblemIndex& ix,
            const ProblemOption& opt, const std::string& rcclId,
            CustomForward<T> customForward, HostAllreduce<T> hostAllreduce,
            HostAllreduce<double> hostAllreduceScalar)
      : customFwd_(std::move(customForward)),
        hostAr_(std::move(hostAllreduce)),
        hostArD_(std::move(hostAllreduceScalar)),
        rank_(opt.rank),
        world_(opt.worldSize),
        ncam_(ix.ncam),
        npt_(ix.npt),
        analytical_(opt.diff == DiffMode::ANALYTICAL),
        implicit_(opt.schur == SchurMode::IMPLICIT),
        lossKind_((int)opt.loss),
        lossD2_((T)(opt.lossDelta * opt.lossDelta)) {
    MEGBA_CHECK(!analytical_ || (CD == 9 && PD == 3 && RD == 2),
                "analytical diff is only available for the BAL (9,3,2) model");
    MEGBA_CHECK(customFwd_ || hasBuiltinResidual(CD, PD, RD),
                "no built-in residual for these dims: provide custom_forward");
    HIP_CHECK(hipSetDevice(opt.deviceIndex));
    HIP_CHECK(hipStreamCreate(&stream_));
    e0_ = ix.split[rank_];
    e1_ = ix.split[rank_ + 1];
    nL_ = e1_ - e0_;
    ptLo_ = ix.ptSplit[rank_];
    ptHi_ = ix.ptSplit[rank_ + 1];
    npL_ = ptHi_ - ptLo_;
    nc_ = (int64_t)ncam_ * CD;
    np_ = (int64_t)npt_ * PD;
    dim_ = nc_ + np_;
    hasInfo_ = !ix.infoSorted.empty();
    {
      double h[3] = {opt.intr[0], opt.intr[1], opt.intr[2]};
      dIntr_ = dalloc<T>(3);
      upCast(dIntr_, h, 3);
    }

    if (world_ > 1) {
      if (rcclId.empty() && hostAr_) {
        // gloo-backed testing fallback (several ranks may share one GPU)
      } else {
        MEGBA_CHECK(rcclId.size() == sizeof(ncclUniqueId),
                    "worldSize>1 requires the RCCL unique id (or a host "
                    "allreduce fallback)");
        ncclUniqueId id;
        std::memcpy(&id, rcclId.data(), sizeof(id));
        RCCL_CHECK(ncclCommInitRank(&comm_, world_, id, rank_));
        hasComm_ = true;
      }
    } else if (getenv("MEGBA_FORCE_RCCL")) {
      // 1-GPU hardening mode: run a real world-1 RCCL communicator so
      // ncclCommInitRank, every ncclAllReduce call site and RCCL-under-
      // hipGraph-capture are exercised on a single GPU (the multi-rank
      // collective SEQUENCE is identical; only the ring is trivial).
      ncclUniqueId id;
      if (rcclId.size() == sizeof(id))
        std::memcpy(&id, rcclId.data(), sizeof(id));
      else
        RCCL_CHECK(ncclGetUniqueId(&id));
      RCCL_CHECK(ncclCommInitRank(&comm_, 1, id, 0));
      hasComm_ = true;
    }

    // Static per-edge data (primary = (pt,cam)-sorted order).
    dCamOf_ = dalloc<int>(nL_);
    dPtOf_ = dalloc<int>(nL_);
    up(dCamOf_, ix.camOf.data() + e0_, nL_);
    up(dPtOf_, ix.ptOf.data() + e0_, nL_);
    if (!prob.camFixed.empty()) {
      dCamFixed_ = dalloc<unsigned char>(ncam_);
      up(dCamFixed_, prob.camFixed.data(), ncam_);
    }
    if (!prob.ptFixed.empty()) {
      dPtFixed_ = dalloc<unsigned char>(npt_);
      up(dPtFixed_, prob.ptFixed.data(), npt_);
    }
    dMeas_ = dalloc<T>(nL_ * RD);
    upCast(dMeas_, ix.measSorted.data() + RD * e0_, nL_ * RD);
    if (customFwd_) {
      dLeaf_ = dalloc<T>(nL_ * GW);
      dMeasSplit_ = dalloc<T>(nL_ * RD);
      std::vector<T> ms(nL_ * RD);
      for (int64_t e = 0; e < nL_; ++e)
        for (int r = 0; r < RD; ++r)
          ms[(int64_t)r * nL_ + e] = (T)ix.measSorted[RD * (e0_ + e) + r];
      up(dMeasSplit_, ms.data(), nL_ * RD);
    }
    if (hasInfo_) {
      dInfo_ = dalloc<T>(nL_ * RW);
      upCast(dInfo_, ix.infoSorted.data() + RW * e0_, nL_ * RW);
    }

    // Parameters (cameras replicated; this rank maintains its point shard).
    dParams_ = dalloc<T>(dim_);
    dParamsBak_ = dalloc<T>(dim_);
    {
      std::vector<T> h(dim_);
      for (int64_t i = 0; i < nc_; ++i) h[i] = (T)prob.cams[i];
      for (int64_t i = 0; i < np_; ++i) h[nc_ + i] = (T)prob.pts[i];
      up(dParams_, h.data(), dim_);
      HIP_CHECK(hipMemcpyAsync(dParamsBak_, dParams_, dim_ * sizeof(T),
                               hipMemcpyDeviceToDevice, stream_));
    }

    // Residual / Jacobian double-buffer (current + accepted).
    for (int s = 0; s < 2; ++s) {
      dR_[s] = dalloc<T>(nL_ * RD);
      dJc_[s] = dalloc<T>(nL_ * CR);
      dJp_[s] = dalloc<T>(nL_ * PR);
    }

    // Linear system (full-size global indexing; only this rank's point
    // ranges of Hll/g_p/temp are maintained).
    dHpp_ = dalloc<T>((int64_t)ncam_ * CC);
    dHll_ = dalloc<T>((int64_t)npt_ * PP);
    dG_ = dalloc<T>(dim_);
    dGBak_ = dalloc<T>(dim_);
    dHppD_ = dalloc<T>((int64_t)ncam_ * CC);
    dHllD_ = dalloc<T>((int64_t)npt_ * PP);
    dHppInv_ = dalloc<T>((int64_t)ncam_ * CC);
    dHllInv_ = dalloc<T>((int64_t)npt_ * PP);
    dDeltaX_ = dalloc<T>(dim_);
    dDeltaXBak_ = dalloc<T>(dim_);
    HIP_CHECK(hipMemsetAsync(dDeltaX_, 0, dim_ * sizeof(T), stream_));
    HIP_CHECK(hipMemsetAsync(dDeltaXBak_, 0, dim_ * sizeof(T), stream_));

    // PCG workspace.
    dP_ = dalloc<T>(nc_);
    dRr_ = dalloc<T>(nc_);
    dZ_ = dalloc<T>(nc_);
    dQ_ = dalloc<T>(nc_);
    dV_ = dalloc<T>(nc_);
    dXBak_ = dalloc<T>(nc_);
    dXBakPrev_ = dalloc<T>(nc_);
    HIP_CHECK(hipMemsetAsync(dXBak_, 0, nc_ * sizeof(T), stream_));
    HIP_CHECK(hipMemsetAsync(dXBakPrev_, 0, nc_ * sizeof(T), stream_));
    dW_ = dalloc<T>(np_);
    dTemp_ = dalloc<T>(np_);
    HIP_CHECK(hipMemsetAsync(dW_, 0, np_ * sizeof(T), stream_));

    // Partial-sum scratch: big enough for both the fixed-shape two-pass
    // reductions (kRedBlocks) and the fused B-apply+dot grid over nc_.
    partCap_ = kRedBlocks > gridFor(nc_) ? kRedBlocks : gridFor(nc_);
    dPart_ = dalloc<double>(partCap_ + 8);
    dPartQ_ = dalloc<double>(partCap_);
    dFail_ = dalloc<int>(2);
    HIP_CHECK(hipHostMalloc((void**)&hScalar_, sizeof(double)));

    // Cam-sorted view of the local edges: slab positions + chunk table.
    {
      std::vector<int> camPos(nL_), rowPtr(ncam_ + 1, 0), ptOfCam(nL_);
      for (int64_t e = 0; e < nL_; ++e) rowPtr[ix.camOf[e0_ + e] + 1]++;
      for (int c = 0; c < ncam_; ++c) rowPtr[c + 1] += rowPtr[c];
      std::vector<int> cursor(rowPtr.begin(), rowPtr.end() - 1);
      for (int64_t e = 0; e < nL_; ++e) {
        const int pos = cursor[ix.camOf[e0_ + e]]++;
        camPos[e] = pos;
        ptOfCam[pos] = ix.ptOf[e0_ + e];
      }
      dCamPos_ = dalloc<int>(nL_);
      dPtOfCam_ = dalloc<int>(nL_);
      up(dCamPos_, camPos.data(), nL_);
      up(dPtOfCam_, ptOfCam.data(), nL_);
      std::vector<int> cCam, cLo, cHi;
      int CHUNK = 256;  // cam-chunk rows (tunable: MEGBA_CHUNK)
      if (const char* c = getenv("MEGBA_CHUNK")) CHUNK = std::atoi(c);
      // Point-band blocking: each camera's slab rows are pt-sorted (the
      // cursor scatter preserves the primary order), so cutting chunks at
      // point-band boundaries and ordering chunks band-major makes every
      // concurrently-resident chunk gather w from the SAME ~2 MB slice —
      // small enough to stay in the XCD L2 instead of being flushed by
      // the streamed J reads (PMC: 78% SQ_WAIT on the unbanded gather,
      // profiles/r02_gather_bands.md).  Band = whole problem for small
      // npt (table unchanged).
      int bandPts = (2 << 20) / (4 * (int)sizeof(T));  // ~2 MB of padded w
      // Occupancy floor: keep >=64 edges (one full wave pass) per
      // (cam, band) chunk on average, or the chunk waves run mostly empty
      // (synth20k measured 694 vs 605 ms/step when 20k cams x 153 bands
      // shattered the table into 16-edge chunks).
      const int64_t occFloor =
          nL_ > 0 ? (int64_t)64 * npt_ * ncam_ / nL_ : 0;
      if (occFloor > bandPts)
        bandPts = (int)std::min<int64_t>(occFloor, npt_);
      if (const char* b = getenv("MEGBA_BAND")) bandPts = std::atoi(b);
      if (bandPts < 1) bandPts = npt_;
      const int nBands = (npt_ + bandPts - 1) / bandPts;
      std::vector<int> cBand;
      for (int c = 0; c < ncam_; ++c) {
        int s = rowPtr[c];
        while (s < rowPtr[c + 1]) {
          const int band = ptOfCam[s] / bandPts;
          const int bandEndPt = (band + 1) * bandPts;
          int e = s;
          const int lim = std::min(s + CHUNK, rowPtr[c + 1]);
          while (e < lim && ptOfCam[e] < bandEndPt) ++e;
          cCam.push_back(c);
          cLo.push_back(s);
          cHi.push_back(e);
          cBand.push_back(band);
          s = e;
        }
      }
      if (nBands > 1) {
        // stable band-major order
        std::vector<int> ord(cCam.size());
        for (size_t i = 0; i < ord.size(); ++i) ord[i] = (int)i;
        std::stable_sort(ord.begin(), ord.end(), [&](int a, int b) {
          return cBand[a] < cBand[b];
        });
        std::vector<int> c2(cCam.size()), l2(cCam.size()), h2(cCam.size());
        for (size_t i = 0; i < ord.size(); ++i) {
          c2[i] = cCam[ord[i]];
          l2[i] = cLo[ord[i]];
          h2[i] = cHi[ord[i]];
        }
        cCam.swap(c2);
        cLo.swap(l2);
        cHi.swap(h2);
      }
      nChunks_ = (int)cCam.size();
      dChCam_ = dalloc<int>(nChunks_);
      dChLo_ = dalloc<int>(nChunks_);
      dChHi_ = dalloc<int>(nChunks_);
      up(dChCam_, cCam.data(), nChunks_);
      up(dChLo_, cLo.data(), nChunks_);
      up(dChHi_, cHi.data(), nChunks_);
    }
    // Run-aligned windows (<=64 edges, whole point runs) for the fused
    // Schur apply; runs longer than 64 edges are flagged and finished by
    // the long-run kernels.
    {
      std::vector<int64_t> wLo, wHi;
      std::vector<unsigned char> wFlag;
      std::vector<int> flagWins, longPts;
      int64_t cur = -1;
      for (int p = ptLo_; p < ptHi_; ++p) {
        const int64_t lo = ix.ptRowPtr[p] - e0_;
        const int64_t hi = ix.ptRowPtr[p + 1] - e0_;
        const int64_t len = hi - lo;
        if (len > 64) {
          longPts.push_back(p);
          for (int64_t ss = lo; ss < hi; ss += 64) {
            flagWins.push_back((int)wLo.size());
            wLo.push_back(ss);
            wHi.push_back(std::min(ss + 64, hi));
            wFlag.push_back(1);
          }
          cur = -1;
          continue;
        }
        if (cur >= 0 && wHi[cur] - wLo[cur] + len <= 64) {
          wHi[cur] = hi;
        } else {
          cur = (int64_t)wLo.size();
          wLo.push_back(lo);
          wHi.push_back(hi);
          wFlag.push_back(0);
        }
      }
      nWin_ = (int)wLo.size();
      nFlagWins_ = (int)flagWins.size();
      nLongPts_ = (int)longPts.size();
      dWinLo_ = dalloc<int64_t>(nWin_);
      dWinHi_ = dalloc<int64_t>(nWin_);
      dWinFlag_ = dalloc<unsigned char>(nWin_);
      up(dWinLo_, wLo.data(), nWin_);
      up(dWinHi_, wHi.data(), nWin_);
      up(dWinFlag_, wFlag.data(), nWin_);
      if (nFlagWins_ > 0) {
        dFlagWins_ = dalloc<int>(nFlagWins_);
        up(dFlagWins_, flagWins.data(), nFlagWins_);
      }
      if (nLongPts_ > 0) {
        dLongPts_ = dalloc<int>(nLongPts_);
        up(dLongPts_, longPts.data(), nLongPts_);
      }
    }
    dSlab_ = dalloc<T>(nL_ * slabWidth());
    dJSlots_ = dalloc<const T*>(3);
    updateJSlots();
    if (implicit_) {
      // packed vector-group layouts (16B groups; padded to a full group)
      constexpr int VEC = 16 / (int)sizeof(T);
      constexpr int NG = (CR + PR + VEC - 1) / VEC;
      dJPk_ = dalloc<T>(nL_ * NG * VEC);
      dJCamPk_ = dalloc<T>(nL_ * NG * VEC);
    }
    {
      constexpr int VEC = 16 / (int)sizeof(T);
      constexpr int XP = (CD + VEC - 1) / VEC * VEC;
      dXPad_ = dalloc<T>((int64_t)ncam_ * XP);
    }
    dWPad_ = dalloc<T>((int64_t)npt_ * 4);
    if (!implicit_) {
      constexpr int VEC = 16 / (int)sizeof(T);
      constexpr int NGH = (CP + VEC - 1) / VEC;
      dHpl_ = dalloc<T>(nL_ * NGH * VEC);
      dHplCam_ = dalloc<T>(nL_ * NGH * VEC);
    }
    sync();
  }

  ~GpuEngine() override {
    if (hScalar_) (void)hipHostFree(hScalar_);
    if (pcgGraphExec_) (void)hipGraphExecDestroy(pcgGraphExec_);
    for (void* p : allocs_) (void)hipFree(p);
    if (hasComm_) (void)ncclCommDestroy(comm_);
    (void)hipStreamDestroy(stream_);
  }

  double forward() override {
    freshCur_ = true;
    if (customFwd_) return forwardCustom();
    zeroScalar();
    if (analytical_) {
      // compile the analytical launch only for its valid dims
      if constexpr (CD == 9 && PD == 3 && RD == 2)
        hipLaunchKernelGGL(kForwardAnalytical<T>, dim3(gridFor(nL_)),
                           dim3(kBlk), 0, stream_, nL_, dCamOf_, dPtOf_,
                           dParams_, ncam_, dMeas_, dCamFixed_, dPtFixed_,
                           dR_[cur_], dJc_[cur_], dJp_[cur_], scalarPtr(),
                           lossKind_, lossD2_);
    } else {
      constexpr int LANES = (GW + 2) / 3;
      bool launched = false;
      if constexpr (CD == 9 && PD == 3 && RD == 2) {
        if (useFwdVS_) {
          hipLaunchKernelGGL(kForwardVS<T>, dim3(gridFor(nL_ * 4)),
                             dim3(kBlk), 0, stream_, nL_, dCamOf_, dPtOf_,
                             dParams_, ncam_, dMeas_, dCamFixed_, dPtFixed_,
                             dR_[cur_], dJc_[cur_], dJp_[cur_], scalarPtr(),
                             lossKind_, lossD2_);
          launched = true;
        }
      }
      if (!launched)
        hipLaunchKernelGGL((kForward<T, CD, PD, RD>),
                           dim3(gridFor(nL_ * LANES)), dim3(kBlk), 0, stream_,
                           nL_, dCamOf_, dPtOf_, dParams_, ncam_, dMeas_,
                           dCamFixed_, dPtFixed_, dR_[cur_], dJc_[cur_],
                           dJp_[cur_], scalarPtr(), lossKind_, lossD2_,
                           dIntr_);
    }
    return globalScalar(ncclSum);
  }

  void acceptForward() override {
    cur_ ^= 1;          // accepted set = dX_[cur_^1]
    freshCur_ = false;  // the (new) current buffers are not yet written
    updateJSlots();
  }

  // Point the device slots at the accepted (bak) buffer set.
  void updateJSlots() {
    const int bak = cur_ ^ 1;
    hipLaunchKernelGGL(kSetPtrSlots<T>, dim3(1), dim3(1), 0, stream_,
                       dJSlots_, dJc_[bak], dJp_[bak], dR_[bak]);
  }

  double forwardCustom() {
    hipLaunchKernelGGL((kGatherLeaves<T, CD, PD>), dim3(gridFor(nL_)),
                       dim3(kBlk), 0, stream_, nL_, dCamOf_, dPtOf_, dParams_,
                       ncam_, dLeaf_);
    sync();  // JetVector ops run on the null stream
    std::vector<JetVec<T>> camL, ptL, ms, res;
    for (int k = 0; k < CD; ++k)
      camL.push_back(jvView<T>(dLeaf_ + (int64_t)k * nL_, nL_, GW, k, true));
    for (int k = 0; k < PD; ++k)
      ptL.push_back(
          jvView<T>(dLeaf_ + (int64_t)(CD + k) * nL_, nL_, GW, CD + k, true));
    for (int r = 0; r < RD; ++r)
      ms.push_back(jvView<T>(dMeasSplit_ + (int64_t)r * nL_, nL_, GW, -1, true));
    customFwd_(camL, ptL, ms, res);
    MEGBA_CHECK((int)res.size() == RD,
                "custom forward must return resDim residuals");
    HIP_CHECK(hipDeviceSynchronize());
    for (int r = 0; r < RD; ++r) {
      MEGBA_CHECK(res[r].kind() == JvKind::DENSE && res[r].nItem == nL_ &&
                      res[r].N == GW && res[r].onGpu,
                  "custom residual must be a dense GPU JetVector "
                  "(N=camDim+ptDim)");
      hipLaunchKernelGGL((kRepackRes<T, CD, PD, RD>), dim3(gridFor(nL_)),
                         dim3(kBlk), 0, stream_, nL_, r, res[r].value->ptr,
                         res[r].grad->ptr, dCamOf_, dPtOf_, dCamFixed_,
                         dPtFixed_, dR_[cur_], dJc_[cur_], dJp_[cur_]);
    }
    if (lossKind_) {
      zeroScalar();
      hipLaunchKernelGGL((kChi2Loss<T, RD>), dim3(gridFor(nL_)), dim3(kBlk),
                         0, stream_, nL_, dR_[cur_], lossKind_, lossD2_,
                         scalarPtr());
    } else {
      reduceDetAsync(dR_[cur_], dR_[cur_], nL_ * RD, ROp::SumSq, scalarPtr());
    }
    return globalScalar(ncclSum);
  }

  void buildLinearSystem() override {
    const int bak = cur_ ^ 1;
    HIP_CHECK(hipMemsetAsync(dHpp_, 0, (int64_t)ncam_ * CC * sizeof(T), stream_));
    HIP_CHECK(hipMemsetAsync(dHll_ + (int64_t)ptLo_ * PP, 0,
                             (int64_t)npL_ * PP * sizeof(T), stream_));
    HIP_CHECK(hipMemsetAsync(dG_, 0, nc_ * sizeof(T), stream_));
    HIP_CHECK(hipMemsetAsync(dG_ + nc_ + (int64_t)ptLo_ * PD, 0,
                             (int64_t)npL_ * PD * sizeof(T), stream_));
    dispatchAssemble(bak);
    if (!implicit_) {
      if (weighted())
        hipLaunchKernelGGL((kFinalizeCam<T, CD, PD, RD, true>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dSlab_, dHplCam_);
      else
        hipLaunchKernelGGL((kFinalizeCam<T, CD, PD, RD, false>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dSlab_, dHplCam_);
    } else {
      if (weighted())
        hipLaunchKernelGGL((kFinalizeCamImpPk<T, CD, PD, RD, true>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dSlab_, dJCamPk_);
      else
        hipLaunchKernelGGL((kFinalizeCamImpPk<T, CD, PD, RD, false>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dSlab_, dJCamPk_);
    }
    // Only the small camera-side quantities cross ranks (the reference
    // allreduced Hpp, Hll AND g, its site A1).
    allreduce(dHpp_, (int64_t)ncam_ * CC, ncclSum);
    allreduce(dG_, nc_, ncclSum);
    sync();
  }

  void backupParams() override {
    HIP_CHECK(hipMemcpyAsync(dParamsBak_, dParams_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
  }
  void rollbackParams() override {
    HIP_CHECK(hipMemcpyAsync(dParams_, dParamsBak_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
  }
  void backupGDx() override {
    HIP_CHECK(hipMemcpyAsync(dDeltaXBak_, dDeltaX_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
    HIP_CHECK(hipMemcpyAsync(dGBak_, dG_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
  }
  void rollbackGDx() override {
    HIP_CHECK(hipMemcpyAsync(dDeltaX_, dDeltaXBak_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
    HIP_CHECK(hipMemcpyAsync(dG_, dGBak_, dim_ * sizeof(T),
                             hipMemcpyDeviceToDevice, stream_));
  }

  void processDiag(double region) override {
    const T f = T(1) + T(1) / (T)region;
    hipLaunchKernelGGL((kDamp<T, CD>), dim3(gridFor((int64_t)ncam_ * CC)),
                       dim3(kBlk), 0, stream_, (int64_t)ncam_ * CC, dHpp_,
                       dHppD_, f, dCamFixed_);
    hipLaunchKernelGGL((kDamp<T, PD>), dim3(gridFor((int64_t)npL_ * PP)),
                       dim3(kBlk), 0, stream_, (int64_t)npL_ * PP,
                       dHll_ + (int64_t)ptLo_ * PP, dHllD_ + (int64_t)ptLo_ * PP,
                       f, dPtFixed_ ? dPtFixed_ + ptLo_ : nullptr);
  }

  int solveLinear(const SolverOptionPCG& opt) override {
    // Block inverses (preconditioner replicated; Cinv on the local shard).
    HIP_CHECK(hipMemsetAsync(dFail_, 0, 2 * sizeof(int), stream_));
    hipLaunchKernelGGL((kInvert<T, CD>), dim3(gridFor(ncam_)), dim3(kBlk), 0,
                       stream_, ncam_, dHppD_, dHppInv_, dFail_);
    hipLaunchKernelGGL((kInvert<T, PD>), dim3(gridFor(npL_)), dim3(kBlk), 0,
                       stream_, npL_, dHllD_ + (int64_t)ptLo_ * PP,
                       dHllInv_ + (int64_t)ptLo_ * PP, dFail_);
    int fail[2] = {0, 0};
    HIP_CHECK(hipMemcpyAsync(fail, dFail_, 2 * sizeof(int),
                             hipMemcpyDeviceToHost, stream_));
    sync();
    if (fail[0]) {
      hipLaunchKernelGGL((kInvertJitter<T, CD>), dim3(gridFor(ncam_)),
                         dim3(kBlk), 0, stream_, ncam_, dHppD_, dHppInv_,
                         dFail_);
      hipLaunchKernelGGL((kInvertJitter<T, PD>), dim3(gridFor(npL_)),
                         dim3(kBlk), 0, stream_, npL_,
                         dHllD_ + (int64_t)ptLo_ * PP,
                         dHllInv_ + (int64_t)ptLo_ * PP, dFail_);
      HIP_CHECK(hipMemcpyAsync(fail, dFail_, 2 * sizeof(int),
                               hipMemcpyDeviceToHost, stream_));
      sync();
      MEGBA_CHECK(!fail[1], "singular Hessian block");
    }

    autoTuneEtx();
    const T* gc = dG_;
    const T* gp = dG_ + nc_;
    // v = gc/world - E Cinv gp  (partial, then the CD*ncam allreduce)
    cinvThenEx(gp, dV_);
    hipLaunchKernelGGL(kVMake<T>, dim3(gridFor(nc_)), dim3(kBlk), 0, stream_,
                       nc_, gc, T(1) / T(world_), dV_);
    allreduce(dV_, nc_, ncclSum);
    // Warm start: x = deltaX camera part (in place in dDeltaX_).
    T* x = dDeltaX_;
    schurApply(x, dQ_);
    hipLaunchKernelGGL(kSub<T>, dim3(gridFor(nc_)), dim3(kBlk), 0, stream_, nc_,
                       dV_, dQ_, dRr_);

    // The loop body is value-uniform (beta/alpha live on-device; the first
    // iteration gets beta = rho/INF = 0 against a zeroed p), so it is
    // captured once as a hipGraph and replayed with ONE host interaction per
    // iteration: the rho readback that drives the reference's refuse/tol
    // control flow.  Refuse semantics are preserved with a two-deep x backup
    // (the body runs speculatively; on refuse x is restored to the value two
    // updates back, exactly what the reference's pre-update check restores).
    HIP_CHECK(hipMemsetAsync(dP_, 0, nc_ * sizeof(T), stream_));
    hipLaunchKernelGGL(kSetScalar, dim3(1), dim3(1), 0, stream_, slotRhoPrev(),
                       INFINITY);
    int n = 0;
    double rho = 0.0, rhoMin = INFINITY;
    bool done = false;
    ensurePcgGraph();
    // Fixed-work mode (tol<=0 with refuse disabled, the bench contract):
    // rho steers nothing, so skip the per-iteration host readback and
    // enqueue all maxIter bodies back-to-back — ONE host sync per solve
    // instead of one per iteration (the N=8 constant-term killer).
    const bool fixedWork = opt.tol <= 0.0 && opt.refuseRatio >= 1e29;
    if (fixedWork) {
      for (; n < opt.maxIter; ++n) {
        if (pcgGraphExec_)
          HIP_CHECK(hipGraphLaunch(pcgGraphExec_, stream_));
        else
          pcgBody();
      }
    } else {
      while (!done && n < opt.maxIter) {
        if (pcgGraphExec_) {
          HIP_CHECK(hipGraphLaunch(pcgGraphExec_, stream_));
        } else {
          pcgBody();
        }
        rho = readScalar(slotRho());
        if (rho > opt.refuseRatio * rhoMin) {
          HIP_CHECK(hipMemcpyAsync(x, dXBakPrev_, nc_ * sizeof(T),
                                   hipMemcpyDeviceToDevice, stream_));
          break;
        }
        rhoMin = rhoMin < rho ? rhoMin : rho;
        ++n;
        done = std::abs(rho) < opt.tol;
      }
    }
    // Back-substitution: deltaX_p = Cinv (g_p - E^T x), fully local.
    spmvEtx(x, dTemp_);
    hipLaunchKernelGGL((kBackSub<T, PD>), dim3(gridFor(npL_)), dim3(kBlk), 0,
                       stream_, npL_, dHllInv_ + (int64_t)ptLo_ * PP,
                       gp + (int64_t)ptLo_ * PD, dTemp_ + (int64_t)ptLo_ * PD,
                       dDeltaX_ + nc_ + (int64_t)ptLo_ * PD);
    sync();
    return n;
  }

  double deltaXL2() override {
    const double camSS = reduceDet(dDeltaX_, dDeltaX_, nc_, ROp::SumSq);
    reduceDetAsync(dDeltaX_ + nc_ + (int64_t)ptLo_ * PD,
                   dDeltaX_ + nc_ + (int64_t)ptLo_ * PD, (int64_t)npL_ * PD,
                   ROp::SumSq, scalarPtr());
    return std::sqrt(camSS + globalScalar(ncclSum));
  }
  double xL2() override {
    const double camSS = reduceDet(dParams_, dParams_, nc_, ROp::SumSq);
    reduceDetAsync(dParams_ + nc_ + (int64_t)ptLo_ * PD,
                   dParams_ + nc_ + (int64_t)ptLo_ * PD, (int64_t)npL_ * PD,
                   ROp::SumSq, scalarPtr());
    return std::sqrt(camSS + globalScalar(ncclSum));
  }
  double gInf() override {
    const double camMax = reduceDet(dG_, dG_, nc_, ROp::AbsMax);
    reduceDetAsync(dG_ + nc_ + (int64_t)ptLo_ * PD,
                   dG_ + nc_ + (int64_t)ptLo_ * PD, (int64_t)npL_ * PD,
                   ROp::AbsMax, scalarPtr());
    const double ptMax = globalScalar(ncclMax);
    return camMax > ptMax ? camMax : ptMax;
  }

  void updateParams() override {
    hipLaunchKernelGGL(kAddAssign<T>, dim3(gridFor(nc_)), dim3(kBlk), 0,
                       stream_, nc_, dDeltaX_, dParams_);
    hipLaunchKernelGGL(kAddAssign<T>, dim3(gridFor((int64_t)npL_ * PD)),
                       dim3(kBlk), 0, stream_, (int64_t)npL_ * PD,
                       dDeltaX_ + nc_ + (int64_t)ptLo_ * PD,
                       dParams_ + nc_ + (int64_t)ptLo_ * PD);
  }

  double rhoDenominator(double chi2Backup) override {
    const int bak = cur_ ^ 1;
    zeroScalar();
    if (implicit_) {
      // packed accepted J + padded deltaX camera vector
      hipLaunchKernelGGL((kPadX<T, CD>), dim3(gridFor(ncam_)), dim3(kBlk),
                         0, stream_, ncam_, dDeltaX_, dXPad_);
      hipLaunchKernelGGL((kRhoDenomPk<T, CD, PD, RD>), dim3(gridFor(nL_)),
                         dim3(kBlk), 0, stream_, nL_, dCamOf_, dPtOf_,
                         dR_[bak], dJPk_, dXPad_, dDeltaX_ + nc_,
                         scalarPtr(), lossKind_, lossD2_);
    } else {
      hipLaunchKernelGGL((kRhoDenom<T, CD, PD, RD>), dim3(gridFor(nL_)),
                         dim3(kBlk), 0, stream_, nL_, dCamOf_, dPtOf_,
                         dR_[bak], dJc_[bak], dJp_[bak], dDeltaX_,
                         dDeltaX_ + nc_, scalarPtr(), lossKind_, lossD2_);
    }
    return globalScalar(ncclSum) - chi2Backup;
  }

  void getParams(double* cams, double* pts) override {
    std::vector<T> hc(nc_);
    HIP_CHECK(hipMemcpy(hc.data(), dParams_, nc_ * sizeof(T),
                        hipMemcpyDeviceToHost));
    for (int64_t i = 0; i < nc_; ++i) cams[i] = (double)hc[i];
    if (world_ > 1 && (hasComm_ || hostAr_)) {
      // point shards: zero the non-local entries, allreduce-sum.
      if (!dPtMerge_) dPtMerge_ = dalloc<T>(np_);
      T* tmp = dPtMerge_;
      HIP_CHECK(hipMemsetAsync(tmp, 0, np_ * sizeof(T), stream_));
      HIP_CHECK(hipMemcpyAsync(tmp + (int64_t)ptLo_ * PD,
                               dParams_ + nc_ + (int64_t)ptLo_ * PD,
                               (int64_t)npL_ * PD * sizeof(T),
                               hipMemcpyDeviceToDevice, stream_));
      allreduce(tmp, np_, ncclSum);
      sync();
      std::vector<T> h(np_);
      HIP_CHECK(hipMemcpy(h.data(), tmp, np_ * sizeof(T), hipMemcpyDeviceToHost));
      for (int64_t i = 0; i < np_; ++i) pts[i] = (double)h[i];
      return;
    }
    std::vector<T> h(np_);
    HIP_CHECK(hipMemcpy(h.data(), dParams_ + nc_, np_ * sizeof(T),
                        hipMemcpyDeviceToHost));
    for (int64_t i = 0; i < np_; ++i) pts[i] = (double)h[i];
  }

  DenseDump dump() const override {
    DenseDump d;
    d.e0 = e0_;
    d.e1 = e1_;
    // r/J of the LAST forward() (survives the acceptForward buffer swap)
    const int fw = freshCur_ ? cur_ : (cur_ ^ 1);
    d.r = down(dR_[fw], nL_ * RD);
    d.Jc = down(dJc_[fw], nL_ * CR);
    d.Jp = down(dJp_[fw], nL_ * PR);
    d.Hpp = down(dHpp_, (int64_t)ncam_ * CC);
    d.Hll = down(dHll_, (int64_t)npt_ * PP);
    if (!implicit_) {
      // device Hpl is packed [group][nL][VEC]; dump as [e][CD][PD]
      constexpr int VEC = 16 / (int)sizeof(T);
      constexpr int NGH = (CP + VEC - 1) / VEC;
      std::vector<double> gm = down(dHpl_, nL_ * NGH * VEC);
      std::vector<double> o((size_t)nL_ * CP);
      for (int64_t e = 0; e < nL_; ++e)
        for (int k = 0; k < CP; ++k)
          o[e * CP + k] =
              gm[((int64_t)(k / VEC) * nL_ + e) * VEC + k % VEC];
      d.Hpl = o;
    }
    d.g = down(dG_, dim_);
    d.deltaX = down(dDeltaX_, dim_);
    // Dump layouts match the CPU engine: r [e][RD], Jc [e][RD][CD],
    // Jp [e][RD][PD] (the device layout is gradient-major; transpose here).
    d.r = transposeR(d.r);
    d.Jc = transposeJ(d.Jc, CD);
    d.Jp = transposeJ(d.Jp, PD);
    return d;
  }

 private:
  bool weighted() const { return hasInfo_ || lossKind_ != 0; }
  int slabWidth() const {
    // must match SlabLayout<CD, PD, RD, EXPL, weighted()>::SW
    return (implicit_ ? 0 : CP) + CR + RD + (weighted() ? CR : 0) +
           (implicit_ ? PR : 0);
  }
  void dispatchAssemble(int bak) {
    auto launch = [&](auto weightedTag, auto explTag) {
      constexpr bool HI = decltype(weightedTag)::value;
      constexpr bool EX = decltype(explTag)::value;
      hipLaunchKernelGGL((kAssembleEdge<T, CD, PD, RD, HI, EX>),
                         dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                         dCamOf_, dPtOf_, dR_[bak], dJc_[bak], dJp_[bak],
                         dInfo_, dHll_, dHpl_, dG_, ncam_, dCamPos_, dSlab_,
                         lossKind_, lossD2_, dJPk_);
      if (nChunks_ > 0) {
        if constexpr (std::is_same<T, double>::value && CD == 9 && PD == 3 &&
                      RD == 2) {
          if (useMfma_) {
            hipLaunchKernelGGL((kAssembleCamMfma<T, CD, PD, RD, HI, EX>),
                               dim3(nChunks_), dim3(64), 0, stream_,
                               nChunks_, dChCam_, dChLo_, dChHi_, dSlab_,
                               dHpp_, dG_);
            return;
          }
        }
        hipLaunchKernelGGL((kAssembleCam<T, CD, PD, RD, HI, EX>),
                           dim3(nChunks_), dim3(128), 0, stream_, nChunks_,
                           dChCam_, dChLo_, dChHi_, dSlab_, dHpp_, dG_);
      }
    };
    using TrueT = std::integral_constant<bool, true>;
    using FalseT = std::integral_constant<bool, false>;
    if (weighted() && !implicit_) launch(TrueT{}, TrueT{});
    else if (weighted() && implicit_) launch(TrueT{}, FalseT{});
    else if (!weighted() && !implicit_) launch(FalseT{}, TrueT{});
    else launch(FalseT{}, FalseT{});
  }
  template <typename U>
  U* dalloc(int64_t n) {
    void* p = nullptr;
    HIP_CHECK(hipMalloc(&p, (n > 0 ? n : 1) * sizeof(U)));
    allocs_.push_back(p);
    return (U*)p;
  }
  template <typename U>
  void up(U* dst, const U* src, int64_t n) {
    if (n > 0)
      HIP_CHECK(hipMemcpyAsync(dst, src, n * sizeof(U), hipMemcpyHostToDevice,
                               stream_));
  }
  void upCast(T* dst, const double* src, int64_t n) {
    std::vector<T> h(n);
    for (int64_t i = 0; i < n; ++i) h[i] = (T)src[i];
    if (n > 0)
      HIP_CHECK(hipMemcpy(dst, h.data(), n * sizeof(T), hipMemcpyHostToDevice));
  }
  std::vector<double> down(const T* src, int64_t n) const {
    std::vector<T> h(n);
    HIP_CHECK(hipMemcpy(h.data(), (void*)src, n * sizeof(T), hipMemcpyDeviceToHost));
    return std::vector<double>(h.begin(), h.end());
  }
  std::vector<double> transposeR(const std::vector<double>& v) const {
    std::vector<double> o(v.size());
    for (int64_t e = 0; e < nL_; ++e)
      for (int row = 0; row < RD; ++row) o[RD * e + row] = v[row * nL_ + e];
    return o;
  }
  std::vector<double> transposeJ(const std::vector<double>& v,
                                 int cols) const {
    std::vector<double> o(v.size());
    for (int64_t e = 0; e < nL_; ++e)
      for (int col = 0; col < cols; ++col)
        for (int row = 0; row < RD; ++row)
          o[e * cols * RD + row * cols + col] =
              v[((int64_t)(col * RD + row)) * nL_ + e];
    return o;
  }
  void sync() { HIP_CHECK(hipStreamSynchronize(stream_)); }
  double* scalarPtr() { return dPart_ + partCap_; }
  double* slotRho() { return dPart_ + partCap_ + 1; }
  double* slotRhoPrev() { return dPart_ + partCap_ + 4; }
  void zeroScalar() {
    HIP_CHECK(hipMemsetAsync(scalarPtr(), 0, sizeof(double), stream_));
  }
  // Read the device scalar accumulator, allreducing across ranks first.
  double globalScalar(ncclRedOp_t op) {
    if (hasComm_) {
      RCCL_CHECK(ncclAllReduce(scalarPtr(), scalarPtr(), 1, ncclDouble, op,
                               comm_, stream_));
      return readScalar(scalarPtr());
    }
    if ((hostArD_ || hostAr_) && world_ > 1) {
      // Host-callback fallback: reduce in full double (the device
      // accumulator already is double) so fp32 multi-rank runs don't lose
      // precision in chi2/deltaXL2/gInf.
      double h = readScalar(scalarPtr());
      if (hostArD_) {
        hostArD_(&h, 1, op == ncclMax ? 'm' : 's');
        return h;
      }
      T v = (T)h;
      hostAr_(&v, 1, op == ncclMax ? 'm' : 's');
      return (double)v;
    }
    return readScalar(scalarPtr());
  }
  double readScalar(double* dptr) {
    // pinned-host destination: the PCG loop does one of these per iteration
    HIP_CHECK(hipMemcpyAsync(hScalar_, dptr, sizeof(double),
                             hipMemcpyDeviceToHost, stream_));
    sync();
    return *hScalar_;
  }
  void allreduce(T* buf, int64_t n, ncclRedOp_t op) {
    if (n == 0 || (world_ == 1 && !hasComm_)) return;
    if (hasComm_) {
      RCCL_CHECK(ncclAllReduce(buf, buf, n,
                               sizeof(T) == 8 ? ncclDouble : ncclFloat, op,
                               comm_, stream_));
      return;
    }
    if (!hostAr_) return;
    // gloo-backed testing fallback: bounce through the host.
    sync();
    std::vector<T> h(n);
    HIP_CHECK(hipMemcpy(h.data(), buf, n * sizeof(T), hipMemcpyDeviceToHost));
    hostAr_(h.data(), n, op == ncclMax ? 'm' : 's');
    HIP_CHECK(hipMemcpy(buf, h.data(), n * sizeof(T), hipMemcpyHostToDevice));
  }
  void reduceDetAsync(const T* a, const T* b, int64_t n, ROp op, double* out) {
    switch (op) {
      case ROp::Dot:
        hipLaunchKernelGGL((kRedPartial<T, ROp::Dot>), dim3(kRedBlocks),
                           dim3(kBlk), 0, stream_, a, b, n, dPart_);
        hipLaunchKernelGGL((kRedFinal<ROp::Dot>), dim3(1), dim3(kBlk), 0,
                           stream_, dPart_, kRedBlocks, out);
        break;
      case ROp::SumSq:
        hipLaunchKernelGGL((kRedPartial<T, ROp::SumSq>), dim3(kRedBlocks),
                           dim3(kBlk), 0, stream_, a, b, n, dPart_);
        hipLaunchKernelGGL((kRedFinal<ROp::SumSq>), dim3(1), dim3(kBlk), 0,
                           stream_, dPart_, kRedBlocks, out);
        break;
      case ROp::AbsMax:
        hipLaunchKernelGGL((kRedPartial<T, ROp::AbsMax>), dim3(kRedBlocks),
                           dim3(kBlk), 0, stream_, a, b, n, dPart_);
        hipLaunchKernelGGL((kRedFinal<ROp::AbsMax>), dim3(1), dim3(kBlk), 0,
                           stream_, dPart_, kRedBlocks, out);
        break;
    }
  }
  double reduceDet(const T* a, const T* b, int64_t n, ROp op) {
    reduceDetAsync(a, b, n, op, scalarPtr());
    return readScalar(scalarPtr());
  }
  template <int D, int MODE>
  void blockMatVec(int nBlk, const T* A, const T* xv, T* yv) {
    hipLaunchKernelGGL((kBlockDiagMatVec<T, D, MODE>),
                       dim3(gridFor((int64_t)nBlk * D)), dim3(kBlk), 0, stream_,
                       nBlk, A, xv, yv);
  }
  // w_local = Cinv * in_local  (pointers offset to the local shard)
  void applyCinv(const T* in, T* out) {
    blockMatVec<PD, 0>(npL_, dHllInv_ + (int64_t)ptLo_ * PP,
                       in + (int64_t)ptLo_ * PD, out + (int64_t)ptLo_ * PD);
  }
  // One-time measured choice between the fused E^T x + Cinv window kernel
  // and the separate-pass pipeline: which wins is problem-dependent
  // (Venice wins fused in BOTH dtypes, final13682-fp32 wins separate —
  // profiles/r02_gather_bands.md Exp 6), so unless an env override is
  // set, time 3 local Schur applies each way on the real data and keep
  // the faster.  Only the LOCAL part runs (no collective), so ranks may
  // even pick differently without breaking lockstep: q partials are
  // per-rank inputs to the allreduce either way.
  void autoTuneEtx() {
    if (etxTuned_) return;
    etxTuned_ = true;
    if (getenv("MEGBA_NO_ETXFUSE") || getenv("MEGBA_ETXFUSE")) return;
    if (nL_ < 200000) return;  // launch-noise regime; keep the default
    hipEvent_t ev[2];
    HIP_CHECK(hipEventCreate(&ev[0]));
    HIP_CHECK(hipEventCreate(&ev[1]));
    float ms[2] = {0, 0};
    for (int variant = 0; variant < 2; ++variant) {
      etxFuse_ = variant == 0;
      // warm-up then timed triple (dQ_/dWPad_/dTemp_ are scratch)
      localSchurEcE(dDeltaX_, dQ_);
      HIP_CHECK(hipEventRecord(ev[0], stream_));
      for (int k = 0; k < 3; ++k) localSchurEcE(dDeltaX_, dQ_);
      HIP_CHECK(hipEventRecord(ev[1], stream_));
      HIP_CHECK(hipEventSynchronize(ev[1]));
      HIP_CHECK(hipEventElapsedTime(&ms[variant], ev[0], ev[1]));
    }
    etxFuse_ = ms[0] <= ms[1];
    (void)hipEventDestroy(ev[0]);
    (void)hipEventDestroy(ev[1]);
  }
  // local (collective-free) E Cinv E^T x into out, current variant
  void localSchurEcE(const T* xv, T* out) {
    if (etxFuse_) {
      etxCinv(xv);
      spmvEx(dWPad_, out);
    } else {
      spmvEtx(xv, dTemp_);
      cinvThenEx(dTemp_, out);
    }
  }

  // w = Cinv in, then out += E w.  w is stored 4-padded so the E-side
  // gather is a single aligned vector load per edge (both modes).
  void cinvThenEx(const T* in, T* out) {
    hipLaunchKernelGGL((kCinvPad<T, PD>), dim3(gridFor(npL_)), dim3(kBlk),
                       0, stream_, npL_, dHllInv_ + (int64_t)ptLo_ * PP,
                       in + (int64_t)ptLo_ * PD,
                       dWPad_ + (int64_t)ptLo_ * 4);
    spmvEx(dWPad_, out);
  }
  void spmvEtx(const T* xv, T* out) {
    hipLaunchKernelGGL(kZeroRange<T>, dim3(gridFor((int64_t)npL_ * PD)),
                       dim3(kBlk), 0, stream_, out + (int64_t)ptLo_ * PD,
                       (int64_t)npL_ * PD);
    if (implicit_) {
      // 16B-aligned padded copy of x: the per-edge camera gather becomes
      // XP/VEC vector loads instead of CD divergent scalar loads.
      hipLaunchKernelGGL((kPadX<T, CD>), dim3(gridFor(ncam_)), dim3(kBlk),
                         0, stream_, ncam_, xv, dXPad_);
      if (etxAtomic_) {
        if (hasInfo_)
          hipLaunchKernelGGL((kSpmvEtxPkAtomic<T, CD, PD, RD, true>),
                             dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                             dCamOf_, dPtOf_, dJPk_,
                             (const T* const*)dJSlots_, dInfo_, lossKind_,
                             lossD2_, dXPad_, out);
        else
          hipLaunchKernelGGL((kSpmvEtxPkAtomic<T, CD, PD, RD, false>),
                             dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                             dCamOf_, dPtOf_, dJPk_,
                             (const T* const*)dJSlots_, (const T*)nullptr,
                             lossKind_, lossD2_, dXPad_, out);
      } else if (hasInfo_)
        hipLaunchKernelGGL((kSpmvEtxPk<T, CD, PD, RD, true>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dCamOf_, dPtOf_, dJPk_,
                           (const T* const*)dJSlots_, dInfo_, lossKind_,
                           lossD2_, dXPad_, out);
      else
        hipLaunchKernelGGL((kSpmvEtxPk<T, CD, PD, RD, false>),
                           dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                           dCamOf_, dPtOf_, dJPk_,
                           (const T* const*)dJSlots_, (const T*)nullptr,
                           lossKind_, lossD2_, dXPad_, out);
    } else {
      hipLaunchKernelGGL((kSpmvEtx<T, CD, PD, RD, false, false>),
                         dim3(gridFor(nL_)), dim3(kBlk), 0, stream_, nL_,
                         dCamOf_, dPtOf_, dHpl_, (const T* const*)nullptr,
                         (const T*)nullptr, 0, T(0), xv, out);
    }
  }
  void spmvEx(const T* wv, T* out) {
    HIP_CHECK(hipMemsetAsync(out, 0, nc_ * sizeof(T), stream_));
    if (implicit_) {
      if (nChunks_ == 0) return;
      hipLaunchKernelGGL((kSpmvExPk<T, CD, PD, RD>), dim3(nChunks_),
                         dim3(64), 0, stream_, nChunks_, dChCam_, dChLo_,
                         dChHi_, dPtOfCam_, dJCamPk_, nL_, wv, out);
      return;
    }
    if (nChunks_ > 0)
      hipLaunchKernelGGL((kSpmvEx<T, CD, PD>), dim3(nChunks_), dim3(64), 0,
                         stream_, nChunks_, dChCam_, dChLo_, dChHi_,
                         dPtOfCam_, dHplCam_, nL_, wv, out);
  }
  // One full PCG iteration (captured as a hipGraph when possible).
  // vs round 1: the p^T q dot partial is fused into the B-apply, the
  // two-deep x backup rotation is fused into kUpdateXR, and the standalone
  // kRedPartial pass + dXBakPrev memcpy are gone (~2 launches + one full
  // read pass over p,q per iteration — part of the N=8 constant term,
  // PLAN_r02 item 3).
  void pcgBody() {
    const int rhoGrid = gridFor(nc_) < kRedBlocks ? gridFor(nc_) : kRedBlocks;
    hipLaunchKernelGGL((kPrecondRho<T, CD>), dim3(rhoGrid), dim3(kBlk), 0,
                       stream_, ncam_, dHppInv_, dRr_, dZ_, dPart_);
    hipLaunchKernelGGL(kXpbySBeta<T>, dim3(gridFor(nc_)), dim3(kBlk), 0,
                       stream_, nc_, dZ_, dPart_, rhoGrid, slotRhoPrev(),
                       slotRho(), dP_);
    schurApply(dP_, dQ_, /*withDot=*/true);
    hipLaunchKernelGGL(kUpdateXRAlpha<T>, dim3(gridFor(nc_)), dim3(kBlk), 0,
                       stream_, nc_, dPart_, rhoGrid, dPartQ_, gridFor(nc_),
                       slotRhoPrev(), dP_, dQ_, dDeltaX_, dXBak_, dXBakPrev_,
                       dRr_);
  }

  // Capture the PCG body as a hipGraph, once.  The implicit path reads the
  // double-buffered accepted J set through device pointer slots (dJSlots_),
  // so accept-time buffer flips never invalidate the graph.  Multi-rank
  // RCCL collectives are captured too (ncclAllReduce is stream-ordered on
  // stream_); any capture failure falls back to the eager path permanently.
  void ensurePcgGraph() {
    if (pcgGraphExec_ || pcgGraphFailed_ || getenv("MEGBA_NO_GRAPH")) return;
    // host-callback collectives (gloo testing fallback) cannot be captured
    if (world_ > 1 && !hasComm_) return;
    sync();
    hipGraph_t graph = nullptr;
    if (hipStreamBeginCapture(stream_, hipStreamCaptureModeThreadLocal) !=
        hipSuccess) {
      pcgGraphFailed_ = true;
      (void)hipGetLastError();
      return;
    }
    bool ok = true;
    try {
      pcgBody();
    } catch (...) {
      ok = false;
    }
    if (hipStreamEndCapture(stream_, &graph) != hipSuccess || !ok || !graph) {
      if (graph) (void)hipGraphDestroy(graph);
      (void)hipGetLastError();
      pcgGraphFailed_ = true;
      return;
    }
    hipGraphExec_t exec = nullptr;
    if (hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0) != hipSuccess) {
      (void)hipGraphDestroy(graph);
      (void)hipGetLastError();
      pcgGraphFailed_ = true;
      return;
    }
    (void)hipGraphDestroy(graph);
    pcgGraphExec_ = exec;
  }

  // out = E Cinv E^T x in one fused pass (see kSchurFused; measured
  // negative, kept env-gated).
  void schurFusedEcE(const T* xv, T* out) {
    HIP_CHECK(hipMemsetAsync(out, 0, nc_ * sizeof(T), stream_));
    hipLaunchKernelGGL((kPadX<T, CD>), dim3(gridFor(ncam_)), dim3(kBlk), 0,
                       stream_, ncam_, xv, dXPad_);
    if (nLongPts_ > 0)
      hipLaunchKernelGGL(kZeroRange<T>, dim3(gridFor((int64_t)npL_ * PD)),
                         dim3(kBlk), 0, stream_, dTemp_ + (int64_t)ptLo_ * PD,
                         (int64_t)npL_ * PD);
    const int grid = (nWin_ + 3) / 4 < 8192 ? (nWin_ + 3) / 4 : 8192;
    auto launch = [&](auto impTag, auto infoTag) {
      constexpr bool IM = decltype(impTag)::value;
      constexpr bool HI = decltype(infoTag)::value;
      hipLaunchKernelGGL((kSchurFused<T, CD, PD, RD, IM, HI>),
                         dim3(grid < 1 ? 1 : grid), dim3(256), 0, stream_,
                         nWin_, dWinLo_, dWinHi_, dWinFlag_, nL_, dCamOf_,
                         dPtOf_, dHpl_, (const T* const*)dJSlots_, dInfo_,
                         lossKind_, lossD2_, dXPad_, dHllInv_, dTemp_, out);
      if (nLongPts_ > 0) {
        hipLaunchKernelGGL((kFusedLongW<T, PD>), dim3(gridFor(nLongPts_)),
                           dim3(kBlk), 0, stream_, nLongPts_, dLongPts_,
                           dHllInv_, dTemp_, dW_, PD);
        const int fg =
            (nFlagWins_ + 3) / 4 < 8192 ? (nFlagWins_ + 3) / 4 : 8192;
        hipLaunchKernelGGL((kFusedLongScatter<T, CD, PD, RD, IM, HI>),
                           dim3(fg < 1 ? 1 : fg), dim3(256), 0, stream_,
                           nFlagWins_, dFlagWins_, dWinLo_, dWinHi_, nL_,
                           dCamOf_, dPtOf_, dHpl_,
                           (const T* const*)dJSlots_, dInfo_, lossKind_,
                           lossD2_, dXPad_, dW_, out);
      }
    };
    using TrueT = std::integral_constant<bool, true>;
    using FalseT = std::integral_constant<bool, false>;
    if (implicit_ && hasInfo_) launch(TrueT{}, TrueT{});
    else if (implicit_) launch(TrueT{}, FalseT{});
    else launch(FalseT{}, FalseT{});
  }

  // w = Cinv E^T x into the 4-padded w vector, one fused window pass
  // (default path; see kEtxCinvFused).
  void etxCinv(const T* xv) {
    hipLaunchKernelGGL((kPadX<T, CD>), dim3(gridFor(ncam_)), dim3(kBlk), 0,
                       stream_, ncam_, xv, dXPad_);
    if (nLongPts_ > 0)
      hipLaunchKernelGGL(kZeroRange<T>, dim3(gridFor((int64_t)npL_ * PD)),
                         dim3(kBlk), 0, stream_, dTemp_ + (int64_t)ptLo_ * PD,
                         (int64_t)npL_ * PD);
    const int grid = (nWin_ + 3) / 4 < 8192 ? (nWin_ + 3) / 4 : 8192;
    auto launch = [&](auto impTag, auto infoTag) {
      constexpr bool IM = decltype(impTag)::value;
      constexpr bool HI = decltype(infoTag)::value;
      hipLaunchKernelGGL((kEtxCinvFused<T, CD, PD, RD, IM, HI>),
                         dim3(grid < 1 ? 1 : grid), dim3(256), 0, stream_,
                         nWin_, dWinLo_, dWinHi_, dWinFlag_, nL_, dCamOf_,
                         dPtOf_, dHpl_, (const T* const*)dJSlots_, dInfo_,
                         lossKind_, lossD2_, dXPad_, dHllInv_, dTemp_,
                         dWPad_);
      if (nLongPts_ > 0)
        hipLaunchKernelGGL((kFusedLongW<T, PD>), dim3(gridFor(nLongPts_)),
                           dim3(kBlk), 0, stream_, nLongPts_, dLongPts_,
                           dHllInv_, dTemp_, dWPad_, 4);
    };
    using TrueT = std::integral_constant<bool, true>;
    using FalseT = std::integral_constant<bool, false>;
    if (implicit_ && hasInfo_) launch(TrueT{}, TrueT{});
    else if (implicit_) launch(TrueT{}, FalseT{});
    else launch(FalseT{}, FalseT{});
  }

  // q = S x = HppD x - E Cinv E^T x  (ONE CD*ncam allreduce; the reference's
  // site A4 needed an additional 3*npt allreduce here).  withDot fuses the
  // x^T q dot partials into the B-apply pass (used by the PCG body).
  void schurApply(const T* xv, T* q, bool withDot = false) {
    if (useFused_) {
      schurFusedEcE(xv, q);
    } else {
      localSchurEcE(xv, q);
    }
    allreduce(q, nc_, ncclSum);
    if (withDot)
      hipLaunchKernelGGL((kBApplyDot<T, CD>), dim3(gridFor(nc_)), dim3(kBlk),
                         0, stream_, ncam_, dHppD_, xv, q, dPartQ_);
    else
      blockMatVec<CD, 1>(ncam_, dHppD_, xv, q);
  }

  hipStream_t stream_{};
  ncclComm_t comm_{};
  bool hasComm_ = false;
  CustomForward<T> customFwd_;
  HostAllreduce<T> hostAr_;
  HostAllreduce<double> hostArD_;
  int rank_, world_, ncam_, npt_;
  int ptLo_ = 0, ptHi_ = 0, npL_ = 0;
  int64_t e0_ = 0, e1_ = 0, nL_ = 0, nc_ = 0, np_ = 0, dim_ = 0;
  bool hasInfo_ = false;
  bool analytical_ = false;
  bool implicit_ = false;
  int lossKind_ = 0;
  T lossD2_ = T(1);
  // MFMA assembly experiment (fp64 BAL only), opt-in via MEGBA_MFMA=1
  bool useMfma_ = getenv("MEGBA_MFMA") != nullptr;
  // Fused one-pass E Cinv E^T apply (opt-in while being measured)
  bool useFused_ = getenv("MEGBA_FUSED") != nullptr;
  // Value-share forward experiment (BAL fp64/fp32), opt-in MEGBA_FWD_VS=1
  bool useFwdVS_ = getenv("MEGBA_FWD_VS") != nullptr;
  // Scan-free E^T x variant, opt-in MEGBA_ETX_ATOMIC=1 while measured
  bool etxAtomic_ = getenv("MEGBA_ETX_ATOMIC") != nullptr;
  // Fused E^T x + Cinv window kernel vs separate passes: auto-tuned on
  // the real data at the first solve (autoTuneEtx); env overrides force.
  bool etxFuse_ = getenv("MEGBA_NO_ETXFUSE") == nullptr;
  bool etxTuned_ = false;
  int nWin_ = 0, nFlagWins_ = 0, nLongPts_ = 0;
  int64_t* dWinLo_{};
  int64_t* dWinHi_{};
  unsigned char* dWinFlag_{};
  int* dFlagWins_{};
  int* dLongPts_{};
  int cur_ = 0;
  bool freshCur_ = false;  // current r/J buffers hold the last forward()
  int nChunks_ = 0;
  int *dCamOf_{}, *dPtOf_{}, *dChCam_{}, *dChLo_{}, *dChHi_{}, *dFail_{};
  int *dCamPos_{}, *dPtOfCam_{};
  T *dMeas_{}, *dInfo_{}, *dLeaf_{}, *dMeasSplit_{}, *dIntr_{};
  unsigned char *dCamFixed_{}, *dPtFixed_{};
  T *dParams_{}, *dParamsBak_{};
  T *dR_[2]{}, *dJc_[2]{}, *dJp_[2]{};
  T *dHpp_{}, *dHll_{}, *dHpl_{}, *dHplCam_{}, *dSlab_{}, *dG_{}, *dGBak_{};
  T *dJPk_{}, *dJCamPk_{};  // implicit: packed [J..] vector groups
  T* dWPad_{};              // 4-padded w for the E-side gather
  T* dXPad_{};              // implicit: XP-padded x for the E^T-side gather
  T *dHppD_{}, *dHllD_{}, *dHppInv_{}, *dHllInv_{};
  T *dDeltaX_{}, *dDeltaXBak_{};
  T *dP_{}, *dRr_{}, *dZ_{}, *dQ_{}, *dV_{}, *dW_{}, *dTemp_{}, *dXBak_{},
      *dXBakPrev_{}, *dPtMerge_{};
  hipGraphExec_t pcgGraphExec_{};
  bool pcgGraphFailed_ = false;
  const T** dJSlots_{};  // device slots: accepted {Jc, Jp, r} pointers
  int partCap_ = kRedBlocks;
  double* dPart_{};
  double* dPartQ_{};
  double* hScalar_{};
  std::vector<void*> allocs_;
};

// Per-dims factory, explicitly instantiated by the gpu_dims_*.hip TUs.
template <typename T, int CD, int PD, int RD>
std::unique_ptr<Engine<T>> makeGpuEngineDims(const BAProblemHost& prob,
                                             const ProblemIndex& ix,
                                             const ProblemOption& opt,
                                             const std::string& rcclId,
                                             CustomForward<T> customForward,
                                             HostAllreduce<T> hostAllreduce,
                                             HostAllreduce<double> hostArD) {
  return std::make_unique<GpuEngine<T, CD, PD, RD>>(
      prob, ix, opt, rcclId, std::move(customForward),
      std::move(hostAllreduce), std::move(hostArD));
}

}  // namespace megba
