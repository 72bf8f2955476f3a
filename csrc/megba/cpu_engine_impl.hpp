// CPU engine implementation, templated over the block dimensions
// <camDim CD, ptDim PD, resDim RD>.  Serves as (a) the BASELINE config-1
// "CPU/Eigen reference path" equivalent, (b) the numerics oracle every HIP
// kernel is tested against, and (c) the world_size>1 correctness testbed.
//
// The reference takes cameraDim/pointDim/resDim as RUNTIME kernel arguments
// (/root/reference/src/edge/build_linear_system.cu:48-146, update.cu:14-41,
// include/common.h:27-46); here they are template parameters instantiated
// over a practical set ({9,6,4} x {3} x {2,3}) so every inner loop still
// unrolls — see cpu_engine.cpp for the dispatch.
#pragma once

#include "cpu_engine.hpp"

#include <algorithm>
#include <cmath>
#include <cstdlib>
#include <cstring>
#ifdef _OPENMP
#include <omp.h>
#endif

#include "analytical.hpp"
#include "bal_functor.hpp"
#include "lm.hpp"
#include "smallmat.hpp"

namespace megba {

template <typename T, int CD, int PD, int RD>
class CpuEngine final : public Engine<T> {
  // Derived block sizes (all loop bounds below are compile-time).
  static constexpr int GW = CD + PD;            // gradient width
  static constexpr int CC = CD * CD;            // Hpp block
  static constexpr int PP = PD * PD;            // Hll block
  static constexpr int CP = CD * PD;            // Hpl block
  static constexpr int CR = CD * RD;            // Jc rows per edge
  static constexpr int PR = PD * RD;            // Jp rows per edge
  static constexpr int RW = RD * (RD + 1) / 2;  // packed sym info entries

 public:
  CpuEngine(const BAProblemHost& prob, const ProblemIndex& ix,
            const ProblemOption& opt, HostAllreduce<T> allreduce,
            CustomForward<T> customForward,
            HostAllreduce<double> allreduceScalar)
      : ar_(std::move(allreduce)),
        arD_(std::move(allreduceScalar)),
        customFwd_(std::move(customForward)),
        rank_(opt.rank),
        world_(opt.worldSize),
        ncam_(ix.ncam),
        npt_(ix.npt),
        analytical_(opt.diff == DiffMode::ANALYTICAL),
        implicit_(opt.schur == SchurMode::IMPLICIT),
        lossKind_((int)opt.loss),
        lossD2_((T)(opt.lossDelta * opt.lossDelta)) {
    for (int i = 0; i < 3; ++i) intr_[i] = (T)opt.intr[i];
    MEGBA_CHECK(!analytical_ || (CD == 9 && PD == 3 && RD == 2),
                "analytical diff is only available for the BAL (9,3,2) model");
    MEGBA_CHECK(customFwd_ || hasBuiltinResidual(CD, PD, RD),
                "no built-in residual for these dims: provide custom_forward");
    e0_ = ix.split[rank_];
    e1_ = ix.split[rank_ + 1];
    nL_ = e1_ - e0_;
#ifdef _OPENMP
    // Clamp the team size to the work size: >64-thread teams on a
    // 256-vCPU host turn the per-region barriers into the dominant cost
    // for small problems (measured 4.5 s/step vs 11 ms for Ladybug-49 on
    // an EPYC 9575F).  An explicit OMP_NUM_THREADS wins.
    if (getenv("OMP_NUM_THREADS") == nullptr) {
      int64_t want = std::max<int64_t>(int64_t(8), nL_ / 4096);
      nt_ = (int)std::min<int64_t>(
          std::min<int64_t>(want, 32), omp_get_max_threads());
    } else {
      nt_ = omp_get_max_threads();
    }
#endif
    camOf_.assign(ix.camOf.begin() + e0_, ix.camOf.begin() + e1_);
    ptOf_.assign(ix.ptOf.begin() + e0_, ix.ptOf.begin() + e1_);
    meas_.resize(nL_ * RD);
    for (int64_t e = 0; e < nL_ * RD; ++e)
      meas_[e] = (T)ix.measSorted[RD * e0_ + e];
    hasInfo_ = !ix.infoSorted.empty();
    if (hasInfo_) {
      info_.resize(nL_ * RW);
      for (int64_t e = 0; e < nL_ * RW; ++e)
        info_[e] = (T)ix.infoSorted[RW * e0_ + e];
    }
    // Local point range (partition is point-aligned: every local edge's
    // point is owned by this rank).
    ptLo_ = ix.ptSplit[rank_];
    ptHi_ = ix.ptSplit[rank_ + 1];
    ptRowPtr_.assign(ix.ptRowPtr.begin(), ix.ptRowPtr.end());

    camFixed_ = prob.camFixed;
    ptFixed_ = prob.ptFixed;
    if (camFixed_.empty()) camFixed_.assign(ncam_, 0);
    if (ptFixed_.empty()) ptFixed_.assign(npt_, 0);
    cams_.resize((size_t)ncam_ * CD);
    pts_.resize((size_t)npt_ * PD);
    for (size_t i = 0; i < cams_.size(); ++i) cams_[i] = (T)prob.cams[i];
    for (size_t i = 0; i < pts_.size(); ++i) pts_[i] = (T)prob.pts[i];
    camsBak_ = cams_;
    ptsBak_ = pts_;

    rCur_.resize(nL_ * RD);
    JcCur_.resize(nL_ * CR);
    JpCur_.resize(nL_ * PR);
    rBak_.resize(nL_ * RD);
    JcBak_.resize(nL_ * CR);
    JpBak_.resize(nL_ * PR);

    Hpp_.assign((size_t)ncam_ * CC, T(0));
    Hll_.assign((size_t)npt_ * PP, T(0));
    Hpl_.assign(implicit_ ? (size_t)0 : (size_t)nL_ * CP, T(0));
    dim_ = (int64_t)ncam_ * CD + (int64_t)npt_ * PD;
    g_.assign(dim_, T(0));
    HppD_.assign(Hpp_.size(), T(0));
    HllD_.assign(Hll_.size(), T(0));
    HppInv_.assign(Hpp_.size(), T(0));
    HllInv_.assign(Hll_.size(), T(0));
    deltaX_.assign(dim_, T(0));
    deltaXBak_.assign(dim_, T(0));
    gBak_.assign(dim_, T(0));
  }

  double forward() override {
    freshCur_ = true;
    if (customFwd_) return forwardCustom();
    using J = Jet<T, GW>;
    T chi2 = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : chi2)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* cp = &cams_[(size_t)camOf_[e] * CD];
      const T* pp = &pts_[(size_t)ptOf_[e] * PD];
      if (analytical_) {
        if constexpr (CD == 9 && PD == 3 && RD == 2) {
          T res[2], jc[2][9], jp[2][3];
          balAnalytical<T>(cp, pp, &meas_[2 * e], res, jc, jp);
          chi2 +=
              lossRho(lossKind_, lossD2_, res[0] * res[0] + res[1] * res[1]);
          for (int row = 0; row < 2; ++row) {
            rCur_[2 * e + row] = res[row];
            for (int i = 0; i < 9; ++i)
              JcCur_[18 * e + 9 * row + i] = jc[row][i];
            for (int i = 0; i < 3; ++i)
              JpCur_[6 * e + 3 * row + i] = jp[row][i];
          }
          zeroFixed(e);
        }
        continue;
      }
      J cam[CD], pt[PD], res[RD];
      for (int i = 0; i < CD; ++i) cam[i] = J::leaf(cp[i], i);
      for (int i = 0; i < PD; ++i) pt[i] = J::leaf(pp[i], CD + i);
      builtinResidual<T, J, CD, PD, RD>(cam, pt, &meas_[RD * e], intr_, res);
      T ss = T(0);
      for (int row = 0; row < RD; ++row) ss += res[row].v * res[row].v;
      chi2 += lossRho(lossKind_, lossD2_, ss);
      for (int row = 0; row < RD; ++row) {
        rCur_[RD * e + row] = res[row].v;
        for (int i = 0; i < CD; ++i)
          JcCur_[CR * e + CD * row + i] = res[row].d[i];
        for (int i = 0; i < PD; ++i)
          JpCur_[PR * e + PD * row + i] = res[row].d[CD + i];
      }
      zeroFixed(e);
    }
    return scalarAr(chi2, 's');
  }

  void buildLinearSystem() override {
    std::fill(Hpp_.begin(), Hpp_.end(), T(0));
    std::fill(Hll_.begin(), Hll_.end(), T(0));
    std::fill(g_.begin(), g_.end(), T(0));
    T* gc = g_.data();
    T* gp = g_.data() + (size_t)ncam_ * CD;

    // Per-edge: weighted J rows, Hpl block, camera blocks into per-thread
    // accumulators (CC + CD values per camera) reduced in fixed order --
    // no atomics, deterministic for a fixed thread count.
    const size_t nAcc = (size_t)ncam_ * (CC + CD);
    if (asmScratch_.size() < (size_t)nt_ * nAcc)
      asmScratch_.assign((size_t)nt_ * nAcc, T(0));
    int team = 1;
#pragma omp parallel num_threads(nt_)
    {
#ifdef _OPENMP
      const int tid = omp_get_thread_num();
#pragma omp single
      team = omp_get_num_threads();
#else
      const int tid = 0;
#endif
      T* accBase = asmScratch_.data() + (size_t)tid * nAcc;
      std::fill(accBase, accBase + nAcc, T(0));
#pragma omp for schedule(static)
      for (int64_t e = 0; e < nL_; ++e) {
        T wJc[RD][CD], wJp[RD][PD], wr[RD];
        weightedRows(e, wJc, wJp, wr);
        if (!implicit_) {
          T* hpl = &Hpl_[CP * e];
          for (int i = 0; i < CD; ++i)
            for (int j = 0; j < PD; ++j) {
              T v = T(0);
              for (int row = 0; row < RD; ++row)
                v += JcBak_[CR * e + CD * row + i] * wJp[row][j];
              hpl[i * PD + j] = v;
            }
        }
        const int c = camOf_[e];
        T* acc = &accBase[(size_t)c * (CC + CD)];
        const T* Jc = &JcBak_[CR * e];
        for (int i = 0; i < CD; ++i) {
          for (int j = 0; j < CD; ++j) {
            T v = T(0);
            for (int row = 0; row < RD; ++row)
              v += Jc[CD * row + i] * wJc[row][j];
            acc[i * CD + j] += v;
          }
          T gv = T(0);
          for (int row = 0; row < RD; ++row) gv += Jc[CD * row + i] * wr[row];
          acc[CC + i] -= gv;
        }
      }
#pragma omp for schedule(static)
      for (int64_t c = 0; c < (int64_t)ncam_; ++c) {
        T* hpp = &Hpp_[(size_t)c * CC];
        T* gcc = &gc[(size_t)c * CD];
        for (int t = 0; t < team; ++t) {
          const T* acc = &asmScratch_[(size_t)t * nAcc + (size_t)c * (CC + CD)];
          for (int i = 0; i < CC; ++i) hpp[i] += acc[i];
          for (int i = 0; i < CD; ++i) gcc[i] += acc[CC + i];
        }
      }
    }

    // Point blocks: local point segments (edges are (pt,cam)-sorted).
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      const int64_t lo = ptRowPtr_[p] - e0_;
      const int64_t hi = ptRowPtr_[p + 1] - e0_;
      T* hll = &Hll_[(size_t)p * PP];
      T* gpt = &gp[(size_t)p * PD];
      for (int64_t e = lo; e < hi; ++e) {
        T wJc[RD][CD], wJp[RD][PD], wr[RD];
        weightedRows(e, wJc, wJp, wr);
        const T* Jp = &JpBak_[PR * e];
        for (int i = 0; i < PD; ++i) {
          for (int j = 0; j < PD; ++j) {
            T v = T(0);
            for (int row = 0; row < RD; ++row)
              v += Jp[PD * row + i] * wJp[row][j];
            hll[i * PD + j] += v;
          }
          T gv = T(0);
          for (int row = 0; row < RD; ++row) gv += Jp[PD * row + i] * wr[row];
          gpt[i] -= gv;
        }
      }
    }

    // Only the small camera-side quantities cross ranks.
    if (ar_) {
      ar_(Hpp_.data(), Hpp_.size(), 's');
      ar_(gc, (size_t)ncam_ * CD, 's');
    }
  }

  void acceptForward() override {
    std::swap(rCur_, rBak_);
    std::swap(JcCur_, JcBak_);
    std::swap(JpCur_, JpBak_);
    // After swap the accepted data is in *Bak_; rhoDenominator reads Bak_,
    // forward overwrites Cur_.
    freshCur_ = false;
  }

  void backupParams() override {
    camsBak_ = cams_;
    ptsBak_ = pts_;
  }
  void rollbackParams() override {
    cams_ = camsBak_;
    pts_ = ptsBak_;
  }
  void backupGDx() override {
    deltaXBak_ = deltaX_;
    gBak_ = g_;
  }
  void rollbackGDx() override {
    deltaX_ = deltaXBak_;
    g_ = gBak_;
  }

  void processDiag(double region) override {
    const T f = T(1) + T(1) / (T)region;
    HppD_ = Hpp_;
    HllD_ = Hll_;
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c) {
      if (camFixed_[c]) {
        for (int i = 0; i < CC; ++i) HppD_[(size_t)c * CC + i] = T(0);
        for (int i = 0; i < CD; ++i)
          HppD_[(size_t)c * CC + i * (CD + 1)] = T(1);
        continue;
      }
      for (int i = 0; i < CD; ++i) HppD_[(size_t)c * CC + i * (CD + 1)] *= f;
    }
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      if (ptFixed_[p]) {
        for (int i = 0; i < PP; ++i) HllD_[(size_t)p * PP + i] = T(0);
        for (int i = 0; i < PD; ++i)
          HllD_[(size_t)p * PP + i * (PD + 1)] = T(1);
        continue;
      }
      for (int i = 0; i < PD; ++i) HllD_[(size_t)p * PP + i * (PD + 1)] *= f;
    }
  }

  int solveLinear(const SolverOptionPCG& opt) override {
    invertBlocks();
    const int64_t nc = (int64_t)ncam_ * CD;
    const T* gc = g_.data();
    const T* gp = g_.data() + nc;
    std::vector<T> w((size_t)npt_ * PD), v(nc), x(nc), r(nc), z(nc), p(nc),
        q(nc), temp((size_t)npt_ * PD), xBak(nc);
    // v = g_c / world - E * Cinv * g_p   (1/world pre-compensates the
    // allreduce of the replicated term; reference schur_pcg_solver.cu:478).
    applyHllInv(gp, w.data());
    spmvEx(w.data(), v.data());
    for (int64_t i = 0; i < nc; ++i) v[i] = gc[i] / (T)world_ - v[i];
    if (ar_) ar_(v.data(), nc, 's');
    // Warm start from current deltaX camera part.
    std::memcpy(x.data(), deltaX_.data(), nc * sizeof(T));
    // r = v - S x
    schurApply(x.data(), q.data(), temp.data(), w.data());
    for (int64_t i = 0; i < nc; ++i) r[i] = v[i] - q[i];

    int n = 0;
    T rho = T(0), rhoPrev = T(0);
    double rhoMin = INFINITY;
    bool done = false;
    while (!done && n < opt.maxIter) {
      applyHppInv(r.data(), z.data());
      rho = dotFull(r.data(), z.data(), nc);
      if ((double)rho > opt.refuseRatio * rhoMin) {
        std::memcpy(x.data(), xBak.data(), nc * sizeof(T));
        break;
      }
      rhoMin = std::min(rhoMin, (double)rho);
      if (n >= 1) {
        const T beta = rhoPrev != T(0) ? rho / rhoPrev : T(0);
        for (int64_t i = 0; i < nc; ++i) p[i] = z[i] + beta * p[i];
      } else {
        std::memcpy(p.data(), z.data(), nc * sizeof(T));
      }
      schurApply(p.data(), q.data(), temp.data(), w.data());
      const T pq = dotFull(p.data(), q.data(), nc);
      const T alpha = pq != T(0) ? rho / pq : T(0);
      std::memcpy(xBak.data(), x.data(), nc * sizeof(T));
      for (int64_t i = 0; i < nc; ++i) {
        x[i] += alpha * p[i];
        r[i] -= alpha * q[i];
      }
      rhoPrev = rho;
      ++n;
      done = std::abs((double)rho) < opt.tol;
    }
    // Back-substitution: deltaX_p = Cinv * (g_p - E^T x).
    spmvEtx(x.data(), temp.data());
    std::memcpy(deltaX_.data(), x.data(), nc * sizeof(T));
    T* dxp = deltaX_.data() + nc;
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int ptI = ptLo_; ptI < ptHi_; ++ptI) {
      T rhs[PD];
      for (int i = 0; i < PD; ++i)
        rhs[i] = gp[PD * ptI + i] - temp[PD * ptI + i];
      matVec<T, PD>(&HllInv_[(size_t)ptI * PP], rhs, &dxp[PD * ptI]);
    }
    return n;
  }

  double deltaXL2() override {
    double s = 0;
    const int64_t nc = (int64_t)ncam_ * CD;
    for (int64_t i = 0; i < nc; ++i) s += (double)deltaX_[i] * deltaX_[i];
    double sp = 0;
    for (int64_t i = nc + (int64_t)ptLo_ * PD; i < nc + (int64_t)ptHi_ * PD;
         ++i)
      sp += (double)deltaX_[i] * deltaX_[i];
    return std::sqrt(s + scalarAr(sp, 's'));
  }
  double xL2() override {
    double s = 0;
    for (const T v : cams_) s += (double)v * v;
    double sp = 0;
    for (int64_t i = (int64_t)ptLo_ * PD; i < (int64_t)ptHi_ * PD; ++i)
      sp += (double)pts_[i] * pts_[i];
    return std::sqrt(s + scalarAr(sp, 's'));
  }
  double gInf() override {
    double m = 0;
    const int64_t nc = (int64_t)ncam_ * CD;
    for (int64_t i = 0; i < nc; ++i)
      m = std::max(m, std::abs((double)g_[i]));
    double mp = 0;
    for (int64_t i = nc + (int64_t)ptLo_ * PD; i < nc + (int64_t)ptHi_ * PD;
         ++i)
      mp = std::max(mp, std::abs((double)g_[i]));
    return std::max(m, scalarAr(mp, 'm'));
  }

  void updateParams() override {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t i = 0; i < (int64_t)cams_.size(); ++i) cams_[i] += deltaX_[i];
    const T* dxp = deltaX_.data() + (size_t)ncam_ * CD;
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t i = (int64_t)ptLo_ * PD; i < (int64_t)ptHi_ * PD; ++i)
      pts_[i] += dxp[i];
  }

  double rhoDenominator(double chi2Backup) override {
    const T* dxc = deltaX_.data();
    const T* dxp = deltaX_.data() + (size_t)ncam_ * CD;
    T s = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : s)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* Jc = &JcBak_[CR * e];
      const T* Jp = &JpBak_[PR * e];
      const T* dc = &dxc[(size_t)camOf_[e] * CD];
      const T* dp = &dxp[(size_t)ptOf_[e] * PD];
      T ss = T(0);
      for (int row = 0; row < RD; ++row) {
        T acc = rBak_[RD * e + row];
        for (int i = 0; i < CD; ++i) acc += Jc[CD * row + i] * dc[i];
        for (int i = 0; i < PD; ++i) acc += Jp[PD * row + i] * dp[i];
        ss += acc * acc;
      }
      s += lossRho(lossKind_, lossD2_, ss);
    }
    return scalarAr(s, 's') - chi2Backup;
  }

  // ---- debug access -------------------------------------------------------
  void getParams(double* cams, double* pts) override {
    for (size_t i = 0; i < cams_.size(); ++i) cams[i] = (double)cams_[i];
    if (ar_ && world_ > 1) {
      // points are sharded: zero non-local entries and sum across ranks
      std::vector<T> full((size_t)npt_ * PD, T(0));
      for (int64_t i = (int64_t)ptLo_ * PD; i < (int64_t)ptHi_ * PD; ++i)
        full[i] = pts_[i];
      ar_(full.data(), full.size(), 's');
      for (size_t i = 0; i < full.size(); ++i) pts[i] = (double)full[i];
      return;
    }
    for (size_t i = 0; i < pts_.size(); ++i) pts[i] = (double)pts_[i];
  }
  DenseDump dump() const override {
    DenseDump d;
    d.e0 = e0_;
    d.e1 = e1_;
    auto cp = [](const std::vector<T>& v) {
      return std::vector<double>(v.begin(), v.end());
    };
    // r/J of the LAST forward() (survives the acceptForward buffer swap)
    d.r = cp(freshCur_ ? rCur_ : rBak_);
    d.Jc = cp(freshCur_ ? JcCur_ : JcBak_);
    d.Jp = cp(freshCur_ ? JpCur_ : JpBak_);
    d.Hpp = cp(Hpp_);
    d.Hll = cp(Hll_);
    d.Hpl = cp(Hpl_);
    d.g = cp(g_);
    d.deltaX = cp(deltaX_);
    return d;
  }

 private:
  double forwardCustom() {
    // Gather the GW parameter leaves + RD measurement rows as JetVectors,
    // run the user expression, repack the residual dual parts.
    std::vector<T> leaf((size_t)GW * nL_), measRow((size_t)RD * nL_);
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* cp = &cams_[(size_t)camOf_[e] * CD];
      const T* pp = &pts_[(size_t)ptOf_[e] * PD];
      for (int k = 0; k < CD; ++k) leaf[(size_t)k * nL_ + e] = cp[k];
      for (int k = 0; k < PD; ++k) leaf[(size_t)(CD + k) * nL_ + e] = pp[k];
      for (int r = 0; r < RD; ++r)
        measRow[(size_t)r * nL_ + e] = meas_[RD * e + r];
    }
    std::vector<JetVec<T>> camL, ptL, ms, res;
    for (int k = 0; k < CD; ++k)
      camL.push_back(jvView<T>(&leaf[(size_t)k * nL_], nL_, GW, k, false));
    for (int k = 0; k < PD; ++k)
      ptL.push_back(
          jvView<T>(&leaf[(size_t)(CD + k) * nL_], nL_, GW, CD + k, false));
    for (int r = 0; r < RD; ++r)
      ms.push_back(jvView<T>(&measRow[(size_t)r * nL_], nL_, GW, -1, false));
    customFwd_(camL, ptL, ms, res);
    MEGBA_CHECK((int)res.size() == RD,
                "custom forward must return resDim residuals");
    for (int r = 0; r < RD; ++r) {
      MEGBA_CHECK(res[r].kind() == JvKind::DENSE && res[r].nItem == nL_ &&
                      res[r].N == GW && !res[r].onGpu,
                  "custom residual must be a dense CPU JetVector (N=camDim+"
                  "ptDim)");
    }
    T chi2 = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : chi2)
    for (int64_t e = 0; e < nL_; ++e) {
      T ss = T(0);
      for (int row = 0; row < RD; ++row) {
        const T v = res[row].value->ptr[e];
        ss += v * v;
        rCur_[RD * e + row] = v;
        const T* g = res[row].grad->ptr;
        for (int k = 0; k < CD; ++k)
          JcCur_[CR * e + CD * row + k] = g[(size_t)k * nL_ + e];
        for (int k = 0; k < PD; ++k)
          JpCur_[PR * e + PD * row + k] = g[(size_t)(CD + k) * nL_ + e];
      }
      chi2 += lossRho(lossKind_, lossD2_, ss);
      zeroFixed(e);
    }
    return scalarAr(chi2, 's');
  }

  // Fixed vertices (g2o parity, reference base_vertex.h `fixed`): their J
  // columns are zeroed after each forward, so all their H blocks and g
  // entries vanish; processDiag then writes an identity diagonal block so
  // the solve is well-posed with deltaX = 0 for them.
  inline void zeroFixed(int64_t e) {
    if (camFixed_[camOf_[e]])
      for (int i = 0; i < CR; ++i) JcCur_[CR * e + i] = T(0);
    if (ptFixed_[ptOf_[e]])
      for (int i = 0; i < PR; ++i) JpCur_[PR * e + i] = T(0);
  }

  // Apply the packed-upper symmetric RDxRD information matrix: out = W in.
  inline void infoApply(const T* wPacked, const T* in, T* out) const {
    for (int i = 0; i < RD; ++i) {
      T s = T(0);
      for (int j = 0; j < RD; ++j) {
        const int a = i < j ? i : j;
        const int b = i < j ? j : i;
        s += wPacked[a * RD - a * (a - 1) / 2 + (b - a)] * in[j];
      }
      out[i] = s;
    }
  }

  // Weighted rows of the ACCEPTED (post-acceptForward) jacobian set.
  // Includes the robust-loss IRLS weight (applied to the weighted side only,
  // so H = sum w J^T W J, g = -sum w J^T W r).
  inline void weightedRows(int64_t e, T wJc[RD][CD], T wJp[RD][PD], T wr[RD]) {
    const T* Jc = &JcBak_[CR * e];
    const T* Jp = &JpBak_[PR * e];
    const T* r = &rBak_[RD * e];
    if (hasInfo_) {
      const T* W = &info_[RW * e];
      T in[RD], out[RD];
      for (int i = 0; i < CD; ++i) {
        for (int row = 0; row < RD; ++row) in[row] = Jc[CD * row + i];
        infoApply(W, in, out);
        for (int row = 0; row < RD; ++row) wJc[row][i] = out[row];
      }
      for (int i = 0; i < PD; ++i) {
        for (int row = 0; row < RD; ++row) in[row] = Jp[PD * row + i];
        infoApply(W, in, out);
        for (int row = 0; row < RD; ++row) wJp[row][i] = out[row];
      }
      infoApply(W, r, wr);
    } else {
      for (int row = 0; row < RD; ++row) {
        for (int i = 0; i < CD; ++i) wJc[row][i] = Jc[CD * row + i];
        for (int i = 0; i < PD; ++i) wJp[row][i] = Jp[PD * row + i];
        wr[row] = r[row];
      }
    }
    if (lossKind_) {
      T ss = T(0);
      for (int row = 0; row < RD; ++row) ss += r[row] * r[row];
      const T w = lossWeight(lossKind_, lossD2_, ss);
      for (int row = 0; row < RD; ++row) {
        for (int i = 0; i < CD; ++i) wJc[row][i] *= w;
        for (int i = 0; i < PD; ++i) wJp[row][i] *= w;
        wr[row] *= w;
      }
    }
  }

  void invertBlocks() {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c) {
      if (!spdInvert<T, CD>(&HppD_[(size_t)c * CC], &HppInv_[(size_t)c * CC]))
        jitterInvert<CD>(&HppD_[(size_t)c * CC], &HppInv_[(size_t)c * CC]);
    }
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      if (!spdInvert<T, PD>(&HllD_[(size_t)p * PP], &HllInv_[(size_t)p * PP]))
        jitterInvert<PD>(&HllD_[(size_t)p * PP], &HllInv_[(size_t)p * PP]);
    }
  }

  template <int D>
  void jitterInvert(const T* a, T* inv) {
    // Numerically semi-definite block: retry with a small relative jitter.
    T buf[D * D];
    T mx = T(0);
    for (int i = 0; i < D; ++i) mx = std::max(mx, std::abs(a[i * D + i]));
    const T eps = (mx > T(0) ? mx : T(1)) * T(1e-10);
    for (int k = 0; k < 40; ++k) {
      const T jit = eps * T(std::pow(10.0, k));
      for (int i = 0; i < D * D; ++i) buf[i] = a[i];
      for (int i = 0; i < D; ++i) buf[i * D + i] += jit;
      if (spdInvert<T, D>(buf, inv)) return;
    }
    MEGBA_CHECK(false, "singular Hessian block");
  }

  // temp[PD*pt] = Hpl^T x over this rank's point segments; fully local (the
  // point side is sharded -- no communication, unlike reference site A4).
  void spmvEtx(const T* x, T* temp) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      const int64_t lo = ptRowPtr_[p] - e0_;
      const int64_t hi = ptRowPtr_[p + 1] - e0_;
      T o[PD];
      for (int j = 0; j < PD; ++j) o[j] = T(0);
      for (int64_t e = lo; e < hi; ++e) {
        const T* xc = &x[(size_t)camOf_[e] * CD];
        if (implicit_) {
          const T* Jc = &JcBak_[CR * e];
          const T* Jp = &JpBak_[PR * e];
          T u[RD];
          for (int row = 0; row < RD; ++row) {
            T s = T(0);
            for (int i = 0; i < CD; ++i) s += Jc[CD * row + i] * xc[i];
            u[row] = s;
          }
          applyInfoLoss(e, u);
          for (int j = 0; j < PD; ++j)
            for (int row = 0; row < RD; ++row)
              o[j] += Jp[PD * row + j] * u[row];
        } else {
          const T* blk = &Hpl_[CP * e];
          for (int j = 0; j < PD; ++j)
            for (int i = 0; i < CD; ++i) o[j] += blk[i * PD + j] * xc[i];
        }
      }
      for (int j = 0; j < PD; ++j) temp[(size_t)p * PD + j] = o[j];
    }
  }

  inline void applyInfoLoss(int64_t e, T u[RD]) {
    if (hasInfo_) {
      T out[RD];
      infoApply(&info_[RW * e], u, out);
      for (int row = 0; row < RD; ++row) u[row] = out[row];
    }
    if (lossKind_) {
      T ss = T(0);
      for (int row = 0; row < RD; ++row) {
        const T rv = rBak_[RD * e + row];
        ss += rv * rv;
      }
      const T w = lossWeight(lossKind_, lossD2_, ss);
      for (int row = 0; row < RD; ++row) u[row] *= w;
    }
  }

  // out[CD*ncam] = partial E w over local edges (caller allreduces CD*ncam
  // -- the ONLY per-iteration collective, 128 KB on Venice).
  void spmvEx(const T* w, T* out) {
    // Per-thread private accumulators + fixed-order tree: the naive
    // per-element "omp atomic" version cost CD contended fp64 RMWs per edge
    // (the CPU PCG's dominant term) and was order-nondeterministic; this is
    // both ~4x faster and bitwise deterministic for a fixed thread count.
    const size_t nCd = (size_t)ncam_ * CD;
    if (exScratch_.size() < (size_t)nt_ * nCd)
      exScratch_.assign((size_t)nt_ * nCd, T(0));
    int team = 1;
#pragma omp parallel num_threads(nt_)
    {
#ifdef _OPENMP
      const int tid = omp_get_thread_num();
#pragma omp single
      team = omp_get_num_threads();
#else
      const int tid = 0;
#endif
      T* acc = exScratch_.data() + (size_t)tid * nCd;
      std::fill(acc, acc + nCd, T(0));
#pragma omp for schedule(static)
      for (int64_t e = 0; e < nL_; ++e) {
        const T* wp = &w[(size_t)ptOf_[e] * PD];
        T* oc = &acc[(size_t)camOf_[e] * CD];
        if (implicit_) {
          const T* Jc = &JcBak_[CR * e];
          const T* Jp = &JpBak_[PR * e];
          T u[RD];
          for (int row = 0; row < RD; ++row) {
            T s = T(0);
            for (int j = 0; j < PD; ++j) s += Jp[PD * row + j] * wp[j];
            u[row] = s;
          }
          applyInfoLoss(e, u);
          for (int i = 0; i < CD; ++i)
            for (int row = 0; row < RD; ++row)
              oc[i] += Jc[CD * row + i] * u[row];
        } else {
          const T* blk = &Hpl_[CP * e];
          for (int i = 0; i < CD; ++i) {
            T s = T(0);
            for (int j = 0; j < PD; ++j) s += blk[i * PD + j] * wp[j];
            oc[i] += s;
          }
        }
      }
#pragma omp for schedule(static)
      for (int64_t i = 0; i < (int64_t)nCd; ++i) {
        T sum = T(0);
        for (int t = 0; t < team; ++t) sum += exScratch_[(size_t)t * nCd + i];
        out[i] = sum;
      }
    }
  }

  void applyHllInv(const T* in, T* out) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p)
      matVec<T, PD>(&HllInv_[(size_t)p * PP], &in[PD * p], &out[PD * p]);
  }
  void applyHppInv(const T* in, T* out) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c)
      matVec<T, CD>(&HppInv_[(size_t)c * CC], &in[CD * c], &out[CD * c]);
  }

  // q = S x = HppD x - E Cinv E^T x   (2 allreduces, reference site A4).
  void schurApply(const T* x, T* q, T* temp, T* w) {
    spmvEtx(x, temp);
    applyHllInv(temp, w);
    spmvEx(w, q);
    if (ar_) ar_(q, (size_t)ncam_ * CD, 's');
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c) {
      T bx[CD];
      matVec<T, CD>(&HppD_[(size_t)c * CC], &x[CD * c], bx);
      for (int i = 0; i < CD; ++i) q[CD * c + i] = bx[i] - q[CD * c + i];
    }
  }

  T dotFull(const T* a, const T* b, int64_t n) const {
    // Replicated vectors: every rank computes the identical full dot, no
    // communication (the reference sliced + host-summed across its devices;
    // with replicated inputs that is redundant).
    T s = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : s)
    for (int64_t i = 0; i < n; ++i) s += a[i] * b[i];
    return s;
  }

  // Reduce a control-flow scalar across ranks in full double precision
  // (falls back to the T-typed callback only if no double variant exists).
  double scalarAr(double v, char op) {
    if (arD_) {
      arD_(&v, 1, op);
      return v;
    }
    if (ar_) {
      T t = (T)v;
      ar_(&t, 1, op);
      return (double)t;
    }
    return v;
  }

  HostAllreduce<T> ar_;
  HostAllreduce<double> arD_;
  CustomForward<T> customFwd_;
  int rank_, world_, ncam_, npt_;
  bool analytical_ = false;
  bool implicit_ = false;
  bool freshCur_ = false;
  int lossKind_ = 0;
  T lossD2_ = T(1);
  T intr_[3] = {T(1), T(0), T(0)};
  int ptLo_ = 0, ptHi_ = 0;
  int64_t e0_ = 0, e1_ = 0, nL_ = 0, dim_ = 0;
  std::vector<int> camOf_, ptOf_;
  std::vector<uint8_t> camFixed_, ptFixed_;
  std::vector<int64_t> ptRowPtr_;
  std::vector<T> meas_, info_;
  bool hasInfo_ = false;
  std::vector<T> cams_, pts_, camsBak_, ptsBak_;
  std::vector<T> rCur_, JcCur_, JpCur_, rBak_, JcBak_, JpBak_;
  std::vector<T> Hpp_, Hll_, Hpl_, g_, HppD_, HllD_, HppInv_, HllInv_;
  std::vector<T> deltaX_, deltaXBak_, gBak_;
  std::vector<T> exScratch_, asmScratch_;  // per-thread reduction buffers
  int nt_ = 1;  // clamped OpenMP team size (see ctor)
};

}  // namespace megba
