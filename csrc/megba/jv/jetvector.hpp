// JetVector: vectorised forward-mode dual numbers over all observations.
//
// Capability parity with the reference's operator layer
// (/root/reference/include/operator/jet_vector.h:22-171 and
// src/operator/jet_vector_math_impl.cu): every scalar in a residual
// expression is a vector over nItem observations with an N-wide dual part.
// Three operand kinds (the reference's 39 hand-written kernel variants are
// generated here from one template over {DENSE, JPV, SCALAR}):
//   DENSE  — value[nItem] + grad[N][nItem] (grad-major, like the reference's
//            grad layout ptr[tid + i*nItem])
//   JPV    — leaf parameter: value only, gradient is the implicit unit
//            vector e_{gradPos} (reference's _gradPosition optimisation)
//   SCALAR — broadcast host scalar (reference's _pureScalarFlag)
// Ops run on GPU (HIP kernels, one thread per item, runtime-N lane loop) or
// CPU (OpenMP).  This layer powers runtime user-defined edges; the built-in
// BAL edge uses the fused register-autodiff kernel instead (gpu_engine.hip),
// which is the performance path.
#pragma once

#include <memory>
#include <vector>

#include "../common.hpp"

namespace megba {

enum class JvKind { DENSE, JPV, SCALAR };

template <typename T>
struct DeviceBuf {  // owns (or views) device or host memory
  T* ptr = nullptr;
  int64_t n = 0;
  bool onGpu = false;
  bool owned = true;
  ~DeviceBuf();
};

template <typename T>
struct JetVec {
  int64_t nItem = 0;
  int N = 0;           // gradient width
  int gradPos = -1;    // >=0 -> JPV
  bool isScalar = false;
  T scalarVal = T(0);
  bool onGpu = false;
  std::shared_ptr<DeviceBuf<T>> value;  // nItem (null if SCALAR)
  std::shared_ptr<DeviceBuf<T>> grad;   // N*nItem grad-major (DENSE only)

  JvKind kind() const {
    return isScalar ? JvKind::SCALAR
                    : (gradPos >= 0 || !grad ? JvKind::JPV : JvKind::DENSE);
  }
};

enum class JvOp { Add, Sub, Mul, Div };
enum class JvUnary { Neg, Abs, Sin, Cos, Sqrt };

// Non-owning view over engine-managed memory (used to hand parameter leaves
// to user forward() callbacks without copies).
template <typename T>
JetVec<T> jvView(T* ptr, int64_t nItem, int N, int gradPos, bool onGpu);

// Return all cached device blocks to the runtime.
void jvPoolTrim();

// Factories -----------------------------------------------------------------
template <typename T>
JetVec<T> jvFromHost(const T* value, const T* grad, int64_t nItem, int N,
                     int gradPos, bool onGpu);
template <typename T>
JetVec<T> jvScalar(T v, int N);
// Download (grad==nullptr skips gradients; JPV/SCALAR materialised densely).
template <typename T>
void jvToHost(const JetVec<T>& a, T* value, T* grad);

// Elementwise ops (output is always DENSE) ----------------------------------
template <typename T>
JetVec<T> jvBinary(JvOp op, const JetVec<T>& a, const JetVec<T>& b);
template <typename T>
JetVec<T> jvUnary(JvUnary op, const JetVec<T>& a);

// Geometry ops (composed from the elementwise layer; reference geo.cuh) -----
// aa[3] -> row-major R[9]
template <typename T>
std::vector<JetVec<T>> jvAngleAxisToRotation(const std::vector<JetVec<T>>& aa);
// wrap theta to (-pi, pi], gradient unchanged (reference SE2
// normalize_rotation2D_Kernel)
template <typename T>
JetVec<T> jvNormalizeAngle(const JetVec<T>& theta);
// theta -> [cos,-sin,sin,cos]
template <typename T>
std::vector<JetVec<T>> jvRotation2D(const JetVec<T>& theta);
// unit quaternion [w,x,y,z] -> row-major R[9]
template <typename T>
std::vector<JetVec<T>> jvQuaternionToRotation(const std::vector<JetVec<T>>& q);
// row-major R[9] -> unit quaternion [w,x,y,z] (per-item Shepperd branch)
template <typename T>
std::vector<JetVec<T>> jvRotationToQuaternion(const std::vector<JetVec<T>>& R);
// q[4] -> normalised q[4]
template <typename T>
std::vector<JetVec<T>> jvNormalizeQuaternion(const std::vector<JetVec<T>>& q);
// p[3] (projected point), intr[3] (f,k1,k2) -> f*(1+k1 r2+k2 r2^2)
template <typename T>
JetVec<T> jvRadialDistortion(const std::vector<JetVec<T>>& p,
                             const std::vector<JetVec<T>>& intr);

}  // namespace megba
