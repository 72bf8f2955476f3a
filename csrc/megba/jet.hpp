// Forward-mode dual numbers ("jets") usable on host and device.
//
// Design (MI355X-first, not a port): the reference (MegBA) vectorises every
// scalar of the residual expression over all edges ("JetVector",
// /root/reference/include/operator/jet_vector.h:22-171) and launches one CUDA
// kernel per elementary op, streaming (N+1)*nItem doubles through HBM per op.
// Here the whole residual is evaluated *in registers* by a single fused HIP
// kernel: each edge's Jet arithmetic lives in VGPRs and only the final
// residual + Jacobian reach memory.  To keep register pressure low on gfx950
// the GPU kernel splits the 12-wide gradient over 4 lanes (Jet<T,3> per lane,
// see gpu/kernels.hip); the CPU oracle uses Jet<T,12>.  Both instantiate the
// same arithmetic below, so CPU and GPU agree to rounding.
#pragma once

#include <cmath>

#ifdef __HIPCC__
#define MEGBA_HD __host__ __device__
#else
#define MEGBA_HD
#endif

namespace megba {

template <typename T, int N>
struct Jet {
  T v;     // value
  T d[N];  // gradient (d res / d param_i for the N seeded directions)

  MEGBA_HD Jet() : v(T(0)) {
    for (int i = 0; i < N; ++i) d[i] = T(0);
  }
  MEGBA_HD explicit Jet(T value) : v(value) {
    for (int i = 0; i < N; ++i) d[i] = T(0);
  }
  MEGBA_HD static Jet leaf(T value, int slot) {
    Jet j(value);
    if (slot >= 0 && slot < N) j.d[slot] = T(1);
    return j;
  }
};

// --- addition -------------------------------------------------------------
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator+(const Jet<T, N>& a, const Jet<T, N>& b) {
  Jet<T, N> r;
  r.v = a.v + b.v;
  for (int i = 0; i < N; ++i) r.d[i] = a.d[i] + b.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator+(const Jet<T, N>& a, T s) {
  Jet<T, N> r = a;
  r.v += s;
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator+(T s, const Jet<T, N>& a) {
  return a + s;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N>& operator+=(Jet<T, N>& a, const Jet<T, N>& b) {
  a.v += b.v;
  for (int i = 0; i < N; ++i) a.d[i] += b.d[i];
  return a;
}

// --- subtraction / negation ----------------------------------------------
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator-(const Jet<T, N>& a, const Jet<T, N>& b) {
  Jet<T, N> r;
  r.v = a.v - b.v;
  for (int i = 0; i < N; ++i) r.d[i] = a.d[i] - b.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator-(const Jet<T, N>& a, T s) {
  Jet<T, N> r = a;
  r.v -= s;
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator-(T s, const Jet<T, N>& a) {
  Jet<T, N> r;
  r.v = s - a.v;
  for (int i = 0; i < N; ++i) r.d[i] = -a.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator-(const Jet<T, N>& a) {
  Jet<T, N> r;
  r.v = -a.v;
  for (int i = 0; i < N; ++i) r.d[i] = -a.d[i];
  return r;
}

// --- multiplication -------------------------------------------------------
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator*(const Jet<T, N>& a, const Jet<T, N>& b) {
  Jet<T, N> r;
  r.v = a.v * b.v;
  for (int i = 0; i < N; ++i) r.d[i] = a.d[i] * b.v + a.v * b.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator*(const Jet<T, N>& a, T s) {
  Jet<T, N> r;
  r.v = a.v * s;
  for (int i = 0; i < N; ++i) r.d[i] = a.d[i] * s;
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator*(T s, const Jet<T, N>& a) {
  return a * s;
}

// --- division -------------------------------------------------------------
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator/(const Jet<T, N>& a, const Jet<T, N>& b) {
  Jet<T, N> r;
  const T inv = T(1) / b.v;
  r.v = a.v * inv;
  // d(a/b) = (da - (a/b) db) / b
  for (int i = 0; i < N; ++i) r.d[i] = (a.d[i] - r.v * b.d[i]) * inv;
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator/(const Jet<T, N>& a, T s) {
  const T inv = T(1) / s;
  return a * inv;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> operator/(T s, const Jet<T, N>& b) {
  Jet<T, N> r;
  const T inv = T(1) / b.v;
  r.v = s * inv;
  const T m = -r.v * inv;
  for (int i = 0; i < N; ++i) r.d[i] = m * b.d[i];
  return r;
}

// --- elementary functions -------------------------------------------------
template <typename T, int N>
MEGBA_HD inline Jet<T, N> sqrt(const Jet<T, N>& a) {
  Jet<T, N> r;
#ifdef __HIP_DEVICE_COMPILE__
  r.v = ::sqrt(a.v);
#else
  r.v = std::sqrt(a.v);
#endif
  const T half_inv = T(0.5) / r.v;
  for (int i = 0; i < N; ++i) r.d[i] = half_inv * a.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> sin(const Jet<T, N>& a) {
  Jet<T, N> r;
#ifdef __HIP_DEVICE_COMPILE__
  const T c = ::cos(a.v);
  r.v = ::sin(a.v);
#else
  const T c = std::cos(a.v);
  r.v = std::sin(a.v);
#endif
  for (int i = 0; i < N; ++i) r.d[i] = c * a.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> cos(const Jet<T, N>& a) {
  Jet<T, N> r;
#ifdef __HIP_DEVICE_COMPILE__
  const T s = ::sin(a.v);
  r.v = ::cos(a.v);
#else
  const T s = std::sin(a.v);
  r.v = std::cos(a.v);
#endif
  for (int i = 0; i < N; ++i) r.d[i] = -s * a.d[i];
  return r;
}
template <typename T, int N>
MEGBA_HD inline Jet<T, N> abs(const Jet<T, N>& a) {
  return a.v < T(0) ? -a : a;
}

// Robust-loss transforms (host + device).  s = squared residual norm.
// rho: the robust cost; w = d rho / d s: the IRLS weight applied to the
// weighted J rows and residual in assembly (H ~ sum w J^T W J).
template <typename T>
MEGBA_HD inline T lossRho(int kind, T d2, T s) {
  if (kind == 1) {  // Huber (delta^2 = d2)
#ifdef __HIP_DEVICE_COMPILE__
    return s <= d2 ? s : T(2) * ::sqrt(d2 * s) - d2;
#else
    return s <= d2 ? s : T(2) * std::sqrt(d2 * s) - d2;
#endif
  }
  if (kind == 2) {  // Cauchy
#ifdef __HIP_DEVICE_COMPILE__
    return d2 * ::log(T(1) + s / d2);
#else
    return d2 * std::log(T(1) + s / d2);
#endif
  }
  return s;
}
template <typename T>
MEGBA_HD inline T lossWeight(int kind, T d2, T s) {
  if (kind == 1) {
#ifdef __HIP_DEVICE_COMPILE__
    return s <= d2 ? T(1) : ::sqrt(d2 / s);
#else
    return s <= d2 ? T(1) : std::sqrt(d2 / s);
#endif
  }
  if (kind == 2) return T(1) / (T(1) + s / d2);
  return T(1);
}

}  // namespace megba
