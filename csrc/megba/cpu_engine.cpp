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

template <typename T>
class CpuEngine final : public Engine<T> {
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
    meas_.resize(nL_ * 2);
    for (int64_t e = 0; e < nL_; ++e) {
      meas_[2 * e] = (T)ix.measSorted[2 * (e0_ + e)];
      meas_[2 * e + 1] = (T)ix.measSorted[2 * (e0_ + e) + 1];
    }
    hasInfo_ = !ix.infoSorted.empty();
    if (hasInfo_) {
      info_.resize(nL_ * 3);
      for (int64_t e = 0; e < nL_ * 3; ++e)
        info_[e] = (T)ix.infoSorted[3 * e0_ + e];
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
    cams_.resize((size_t)ncam_ * 9);
    pts_.resize((size_t)npt_ * 3);
    for (size_t i = 0; i < cams_.size(); ++i) cams_[i] = (T)prob.cams[i];
    for (size_t i = 0; i < pts_.size(); ++i) pts_[i] = (T)prob.pts[i];
    camsBak_ = cams_;
    ptsBak_ = pts_;

    rCur_.resize(nL_ * 2);
    JcCur_.resize(nL_ * 18);
    JpCur_.resize(nL_ * 6);
    rBak_.resize(nL_ * 2);
    JcBak_.resize(nL_ * 18);
    JpBak_.resize(nL_ * 6);

    Hpp_.assign((size_t)ncam_ * 81, T(0));
    Hll_.assign((size_t)npt_ * 9, T(0));
    Hpl_.assign(implicit_ ? (size_t)0 : (size_t)nL_ * 27, T(0));
    dim_ = (int64_t)ncam_ * 9 + (int64_t)npt_ * 3;
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
    using J = Jet<T, 12>;
    T chi2 = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : chi2)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* cp = &cams_[(size_t)camOf_[e] * 9];
      const T* pp = &pts_[(size_t)ptOf_[e] * 3];
      if (analytical_) {
        T res[2], jc[2][9], jp[2][3];
        balAnalytical<T>(cp, pp, &meas_[2 * e], res, jc, jp);
        chi2 += lossRho(lossKind_, lossD2_, res[0] * res[0] + res[1] * res[1]);
        for (int row = 0; row < 2; ++row) {
          rCur_[2 * e + row] = res[row];
          for (int i = 0; i < 9; ++i) JcCur_[18 * e + 9 * row + i] = jc[row][i];
          for (int i = 0; i < 3; ++i) JpCur_[6 * e + 3 * row + i] = jp[row][i];
        }
        zeroFixed(e);
        continue;
      }
      J cam[9], pt[3], res[2];
      for (int i = 0; i < 9; ++i) cam[i] = J::leaf(cp[i], i);
      for (int i = 0; i < 3; ++i) pt[i] = J::leaf(pp[i], 9 + i);
      balReprojectionError<T, J>(cam, pt, &meas_[2 * e], res);
      chi2 += lossRho(lossKind_, lossD2_,
                      res[0].v * res[0].v + res[1].v * res[1].v);
      for (int row = 0; row < 2; ++row) {
        rCur_[2 * e + row] = res[row].v;
        for (int i = 0; i < 9; ++i) JcCur_[18 * e + 9 * row + i] = res[row].d[i];
        for (int i = 0; i < 3; ++i) JpCur_[6 * e + 3 * row + i] = res[row].d[9 + i];
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
    T* gp = g_.data() + (size_t)ncam_ * 9;

    // Per-edge: weighted J rows, Hpl block, camera blocks into per-thread
    // accumulators (90 values per camera: 81 Hpp + 9 g) reduced in fixed
    // order -- no atomics, deterministic for a fixed thread count.
    const size_t n90 = (size_t)ncam_ * 90;
    if (asmScratch_.size() < (size_t)nt_ * n90)
      asmScratch_.assign((size_t)nt_ * n90, T(0));
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
      T* accBase = asmScratch_.data() + (size_t)tid * n90;
      std::fill(accBase, accBase + n90, T(0));
#pragma omp for schedule(static)
      for (int64_t e = 0; e < nL_; ++e) {
        T wJc[2][9], wJp[2][3], wr[2];
        weightedRows(e, wJc, wJp, wr);
        if (!implicit_) {
          T* hpl = &Hpl_[27 * e];
          for (int i = 0; i < 9; ++i)
            for (int j = 0; j < 3; ++j)
              hpl[i * 3 + j] = JcBak_[18 * e + i] * wJp[0][j] +
                               JcBak_[18 * e + 9 + i] * wJp[1][j];
        }
        const int c = camOf_[e];
        T* acc = &accBase[(size_t)c * 90];
        const T* Jc = &JcBak_[18 * e];
        for (int i = 0; i < 9; ++i) {
          for (int j = 0; j < 9; ++j)
            acc[i * 9 + j] += Jc[i] * wJc[0][j] + Jc[9 + i] * wJc[1][j];
          acc[81 + i] -= Jc[i] * wr[0] + Jc[9 + i] * wr[1];
        }
      }
#pragma omp for schedule(static)
      for (int64_t c = 0; c < (int64_t)ncam_; ++c) {
        T* hpp = &Hpp_[(size_t)c * 81];
        T* gcc = &gc[(size_t)c * 9];
        for (int t = 0; t < team; ++t) {
          const T* acc = &asmScratch_[(size_t)t * n90 + (size_t)c * 90];
          for (int i = 0; i < 81; ++i) hpp[i] += acc[i];
          for (int i = 0; i < 9; ++i) gcc[i] += acc[81 + i];
        }
      }
    }

    // Point blocks: local point segments (edges are (pt,cam)-sorted).
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      const int64_t lo = ptRowPtr_[p] - e0_;
      const int64_t hi = ptRowPtr_[p + 1] - e0_;
      T* hll = &Hll_[(size_t)p * 9];
      T* gpt = &gp[(size_t)p * 3];
      for (int64_t e = lo; e < hi; ++e) {
        T wJc[2][9], wJp[2][3], wr[2];
        weightedRows(e, wJc, wJp, wr);
        const T* Jp = &JpBak_[6 * e];
        for (int i = 0; i < 3; ++i) {
          for (int j = 0; j < 3; ++j)
            hll[i * 3 + j] += Jp[i] * wJp[0][j] + Jp[3 + i] * wJp[1][j];
          gpt[i] -= Jp[i] * wr[0] + Jp[3 + i] * wr[1];
        }
      }
    }

    // Only the small camera-side quantities cross ranks.
    if (ar_) {
      ar_(Hpp_.data(), Hpp_.size(), 's');
      ar_(gc, (size_t)ncam_ * 9, 's');
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
        for (int i = 0; i < 81; ++i) HppD_[(size_t)c * 81 + i] = T(0);
        for (int i = 0; i < 9; ++i) HppD_[(size_t)c * 81 + i * 10] = T(1);
        continue;
      }
      for (int i = 0; i < 9; ++i) HppD_[(size_t)c * 81 + i * 10] *= f;
    }
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      if (ptFixed_[p]) {
        for (int i = 0; i < 9; ++i) HllD_[(size_t)p * 9 + i] = T(0);
        for (int i = 0; i < 3; ++i) HllD_[(size_t)p * 9 + i * 4] = T(1);
        continue;
      }
      for (int i = 0; i < 3; ++i) HllD_[(size_t)p * 9 + i * 4] *= f;
    }
  }

  int solveLinear(const SolverOptionPCG& opt) override {
    invertBlocks();
    const int64_t nc = (int64_t)ncam_ * 9;
    const T* gc = g_.data();
    const T* gp = g_.data() + nc;
    std::vector<T> w((size_t)npt_ * 3), v(nc), x(nc), r(nc), z(nc), p(nc),
        q(nc), temp((size_t)npt_ * 3), xBak(nc);
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
        const T beta = rho / rhoPrev;
        for (int64_t i = 0; i < nc; ++i) p[i] = z[i] + beta * p[i];
      } else {
        std::memcpy(p.data(), z.data(), nc * sizeof(T));
      }
      schurApply(p.data(), q.data(), temp.data(), w.data());
      const T pq = dotFull(p.data(), q.data(), nc);
      const T alpha = rho / pq;
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
      T rhs[3];
      for (int i = 0; i < 3; ++i) rhs[i] = gp[3 * ptI + i] - temp[3 * ptI + i];
      matVec<T, 3>(&HllInv_[(size_t)ptI * 9], rhs, &dxp[3 * ptI]);
    }
    return n;
  }

  double deltaXL2() override {
    double s = 0;
    const int64_t nc = (int64_t)ncam_ * 9;
    for (int64_t i = 0; i < nc; ++i) s += (double)deltaX_[i] * deltaX_[i];
    double sp = 0;
    for (int64_t i = nc + (int64_t)ptLo_ * 3; i < nc + (int64_t)ptHi_ * 3; ++i)
      sp += (double)deltaX_[i] * deltaX_[i];
    return std::sqrt(s + scalarAr(sp, 's'));
  }
  double xL2() override {
    double s = 0;
    for (const T v : cams_) s += (double)v * v;
    double sp = 0;
    for (int64_t i = (int64_t)ptLo_ * 3; i < (int64_t)ptHi_ * 3; ++i)
      sp += (double)pts_[i] * pts_[i];
    return std::sqrt(s + scalarAr(sp, 's'));
  }
  double gInf() override {
    double m = 0;
    const int64_t nc = (int64_t)ncam_ * 9;
    for (int64_t i = 0; i < nc; ++i)
      m = std::max(m, std::abs((double)g_[i]));
    double mp = 0;
    for (int64_t i = nc + (int64_t)ptLo_ * 3; i < nc + (int64_t)ptHi_ * 3; ++i)
      mp = std::max(mp, std::abs((double)g_[i]));
    return std::max(m, scalarAr(mp, 'm'));
  }

  void updateParams() override {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t i = 0; i < (int64_t)cams_.size(); ++i) cams_[i] += deltaX_[i];
    const T* dxp = deltaX_.data() + (size_t)ncam_ * 9;
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t i = (int64_t)ptLo_ * 3; i < (int64_t)ptHi_ * 3; ++i)
      pts_[i] += dxp[i];
  }

  double rhoDenominator(double chi2Backup) override {
    const T* dxc = deltaX_.data();
    const T* dxp = deltaX_.data() + (size_t)ncam_ * 9;
    T s = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : s)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* Jc = &JcBak_[18 * e];
      const T* Jp = &JpBak_[6 * e];
      const T* dc = &dxc[(size_t)camOf_[e] * 9];
      const T* dp = &dxp[(size_t)ptOf_[e] * 3];
      T acc2[2];
      for (int row = 0; row < 2; ++row) {
        T acc = rBak_[2 * e + row];
        for (int i = 0; i < 9; ++i) acc += Jc[9 * row + i] * dc[i];
        for (int i = 0; i < 3; ++i) acc += Jp[3 * row + i] * dp[i];
        acc2[row] = acc;
      }
      s += lossRho(lossKind_, lossD2_,
                   acc2[0] * acc2[0] + acc2[1] * acc2[1]);
    }
    return scalarAr(s, 's') - chi2Backup;
  }

  // ---- debug access -------------------------------------------------------
  void getParams(double* cams, double* pts) override {
    for (size_t i = 0; i < cams_.size(); ++i) cams[i] = (double)cams_[i];
    if (ar_ && world_ > 1) {
      // points are sharded: zero non-local entries and sum across ranks
      std::vector<T> full((size_t)npt_ * 3, T(0));
      for (int64_t i = (int64_t)ptLo_ * 3; i < (int64_t)ptHi_ * 3; ++i)
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
    // Gather the 12 parameter leaves + 2 measurement rows as JetVectors,
    // run the user expression, repack the residual dual parts.
    std::vector<T> leaf((size_t)12 * nL_), measRow((size_t)2 * nL_);
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int64_t e = 0; e < nL_; ++e) {
      const T* cp = &cams_[(size_t)camOf_[e] * 9];
      const T* pp = &pts_[(size_t)ptOf_[e] * 3];
      for (int k = 0; k < 9; ++k) leaf[(size_t)k * nL_ + e] = cp[k];
      for (int k = 0; k < 3; ++k) leaf[(size_t)(9 + k) * nL_ + e] = pp[k];
      measRow[e] = meas_[2 * e];
      measRow[nL_ + e] = meas_[2 * e + 1];
    }
    std::vector<JetVec<T>> camL, ptL, ms, res;
    for (int k = 0; k < 9; ++k)
      camL.push_back(jvView<T>(&leaf[(size_t)k * nL_], nL_, 12, k, false));
    for (int k = 0; k < 3; ++k)
      ptL.push_back(jvView<T>(&leaf[(size_t)(9 + k) * nL_], nL_, 12, 9 + k, false));
    for (int r = 0; r < 2; ++r)
      ms.push_back(jvView<T>(&measRow[(size_t)r * nL_], nL_, 12, -1, false));
    customFwd_(camL, ptL, ms, res);
    MEGBA_CHECK(res.size() == 2, "custom forward must return 2 residuals");
    for (int r = 0; r < 2; ++r) {
      MEGBA_CHECK(res[r].kind() == JvKind::DENSE && res[r].nItem == nL_ &&
                      res[r].N == 12 && !res[r].onGpu,
                  "custom residual must be a dense CPU JetVector (N=12)");
    }
    T chi2 = T(0);
#pragma omp parallel for num_threads(nt_) schedule(static) reduction(+ : chi2)
    for (int64_t e = 0; e < nL_; ++e) {
      const T v0 = res[0].value->ptr[e];
      const T v1 = res[1].value->ptr[e];
      chi2 += lossRho(lossKind_, lossD2_, v0 * v0 + v1 * v1);
      for (int row = 0; row < 2; ++row) {
        const T v = res[row].value->ptr[e];
        rCur_[2 * e + row] = v;
        const T* g = res[row].grad->ptr;
        for (int k = 0; k < 9; ++k)
          JcCur_[18 * e + 9 * row + k] = g[(size_t)k * nL_ + e];
        for (int k = 0; k < 3; ++k)
          JpCur_[6 * e + 3 * row + k] = g[(size_t)(9 + k) * nL_ + e];
      }
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
      for (int i = 0; i < 18; ++i) JcCur_[18 * e + i] = T(0);
    if (ptFixed_[ptOf_[e]])
      for (int i = 0; i < 6; ++i) JpCur_[6 * e + i] = T(0);
  }

  // Weighted rows of the ACCEPTED (post-acceptForward) jacobian set.
  // Includes the robust-loss IRLS weight (applied to the weighted side only,
  // so H = sum w J^T W J, g = -sum w J^T W r).
  inline void weightedRows(int64_t e, T wJc[2][9], T wJp[2][3], T wr[2]) {
    const T* Jc = &JcBak_[18 * e];
    const T* Jp = &JpBak_[6 * e];
    const T* r = &rBak_[2 * e];
    if (hasInfo_) {
      const T w00 = info_[3 * e], w01 = info_[3 * e + 1], w11 = info_[3 * e + 2];
      for (int i = 0; i < 9; ++i) {
        wJc[0][i] = w00 * Jc[i] + w01 * Jc[9 + i];
        wJc[1][i] = w01 * Jc[i] + w11 * Jc[9 + i];
      }
      for (int i = 0; i < 3; ++i) {
        wJp[0][i] = w00 * Jp[i] + w01 * Jp[3 + i];
        wJp[1][i] = w01 * Jp[i] + w11 * Jp[3 + i];
      }
      wr[0] = w00 * r[0] + w01 * r[1];
      wr[1] = w01 * r[0] + w11 * r[1];
    } else {
      for (int i = 0; i < 9; ++i) {
        wJc[0][i] = Jc[i];
        wJc[1][i] = Jc[9 + i];
      }
      for (int i = 0; i < 3; ++i) {
        wJp[0][i] = Jp[i];
        wJp[1][i] = Jp[3 + i];
      }
      wr[0] = r[0];
      wr[1] = r[1];
    }
    if (lossKind_) {
      const T w =
          lossWeight(lossKind_, lossD2_, r[0] * r[0] + r[1] * r[1]);
      for (int i = 0; i < 9; ++i) {
        wJc[0][i] *= w;
        wJc[1][i] *= w;
      }
      for (int i = 0; i < 3; ++i) {
        wJp[0][i] *= w;
        wJp[1][i] *= w;
      }
      wr[0] *= w;
      wr[1] *= w;
    }
  }

  void invertBlocks() {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c) {
      if (!spdInvert<T, 9>(&HppD_[(size_t)c * 81], &HppInv_[(size_t)c * 81]))
        jitterInvert<9>(&HppD_[(size_t)c * 81], &HppInv_[(size_t)c * 81]);
    }
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      if (!spdInvert<T, 3>(&HllD_[(size_t)p * 9], &HllInv_[(size_t)p * 9]))
        jitterInvert<3>(&HllD_[(size_t)p * 9], &HllInv_[(size_t)p * 9]);
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

  // temp[3*pt] = Hpl^T x over this rank's point segments; fully local (the
  // point side is sharded -- no communication, unlike reference site A4).
  void spmvEtx(const T* x, T* temp) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p) {
      const int64_t lo = ptRowPtr_[p] - e0_;
      const int64_t hi = ptRowPtr_[p + 1] - e0_;
      T o[3] = {T(0), T(0), T(0)};
      for (int64_t e = lo; e < hi; ++e) {
        const T* xc = &x[(size_t)camOf_[e] * 9];
        if (implicit_) {
          const T* Jc = &JcBak_[18 * e];
          const T* Jp = &JpBak_[6 * e];
          T u0 = T(0), u1 = T(0);
          for (int i = 0; i < 9; ++i) {
            u0 += Jc[i] * xc[i];
            u1 += Jc[9 + i] * xc[i];
          }
          applyInfo(e, u0, u1);
          for (int j = 0; j < 3; ++j) o[j] += Jp[j] * u0 + Jp[3 + j] * u1;
        } else {
          const T* blk = &Hpl_[27 * e];
          for (int j = 0; j < 3; ++j)
            for (int i = 0; i < 9; ++i) o[j] += blk[i * 3 + j] * xc[i];
        }
      }
      for (int j = 0; j < 3; ++j) temp[(size_t)p * 3 + j] = o[j];
    }
  }

  inline void applyInfo(int64_t e, T& u0, T& u1) {
    if (hasInfo_) {
      const T w00 = info_[3 * e], w01 = info_[3 * e + 1],
              w11 = info_[3 * e + 2];
      const T a = w00 * u0 + w01 * u1;
      u1 = w01 * u0 + w11 * u1;
      u0 = a;
    }
    if (lossKind_) {
      const T r0 = rBak_[2 * e], r1 = rBak_[2 * e + 1];
      const T w = lossWeight(lossKind_, lossD2_, r0 * r0 + r1 * r1);
      u0 *= w;
      u1 *= w;
    }
  }

  // out[9ncam] = partial E w over local edges (caller allreduces 9*ncam --
  // the ONLY per-iteration collective, 128 KB on Venice).
  void spmvEx(const T* w, T* out) {
    // Per-thread private accumulators + fixed-order tree: the naive
    // per-element "omp atomic" version cost 9 contended fp64 RMWs per edge
    // (the CPU PCG's dominant term) and was order-nondeterministic; this is
    // both ~4x faster and bitwise deterministic for a fixed thread count.
    const size_t n9 = (size_t)ncam_ * 9;
    if (exScratch_.size() < (size_t)nt_ * n9)
      exScratch_.assign((size_t)nt_ * n9, T(0));
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
      T* acc = exScratch_.data() + (size_t)tid * n9;
      std::fill(acc, acc + n9, T(0));
#pragma omp for schedule(static)
      for (int64_t e = 0; e < nL_; ++e) {
        const T* wp = &w[(size_t)ptOf_[e] * 3];
        T* oc = &acc[(size_t)camOf_[e] * 9];
        if (implicit_) {
          const T* Jc = &JcBak_[18 * e];
          const T* Jp = &JpBak_[6 * e];
          T u0 = Jp[0] * wp[0] + Jp[1] * wp[1] + Jp[2] * wp[2];
          T u1 = Jp[3] * wp[0] + Jp[4] * wp[1] + Jp[5] * wp[2];
          applyInfo(e, u0, u1);
          for (int i = 0; i < 9; ++i) oc[i] += Jc[i] * u0 + Jc[9 + i] * u1;
        } else {
          const T* blk = &Hpl_[27 * e];
          for (int i = 0; i < 9; ++i)
            oc[i] += blk[i * 3] * wp[0] + blk[i * 3 + 1] * wp[1] +
                     blk[i * 3 + 2] * wp[2];
        }
      }
#pragma omp for schedule(static)
      for (int64_t i = 0; i < (int64_t)n9; ++i) {
        T sum = T(0);
        for (int t = 0; t < team; ++t) sum += exScratch_[(size_t)t * n9 + i];
        out[i] = sum;
      }
    }
  }

  void applyHllInv(const T* in, T* out) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int p = ptLo_; p < ptHi_; ++p)
      matVec<T, 3>(&HllInv_[(size_t)p * 9], &in[3 * p], &out[3 * p]);
  }
  void applyHppInv(const T* in, T* out) {
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c)
      matVec<T, 9>(&HppInv_[(size_t)c * 81], &in[9 * c], &out[9 * c]);
  }

  // q = S x = HppD x - E Cinv E^T x   (2 allreduces, reference site A4).
  void schurApply(const T* x, T* q, T* temp, T* w) {
    spmvEtx(x, temp);
    applyHllInv(temp, w);
    spmvEx(w, q);
    if (ar_) ar_(q, (size_t)ncam_ * 9, 's');
#pragma omp parallel for num_threads(nt_) schedule(static)
    for (int c = 0; c < ncam_; ++c) {
      T bx[9];
      matVec<T, 9>(&HppD_[(size_t)c * 81], &x[9 * c], bx);
      for (int i = 0; i < 9; ++i) q[9 * c + i] = bx[i] - q[9 * c + i];
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

template <typename T>
std::unique_ptr<Engine<T>> makeCpuEngine(const BAProblemHost& prob,
                                         const ProblemIndex& ix,
                                         const ProblemOption& opt,
                                         HostAllreduce<T> allreduce,
                                         CustomForward<T> customForward,
                                         HostAllreduce<double> allreduceScalar) {
  return std::make_unique<CpuEngine<T>>(prob, ix, opt, std::move(allreduce),
                                        std::move(customForward),
                                        std::move(allreduceScalar));
}

template std::unique_ptr<Engine<double>> makeCpuEngine<double>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    HostAllreduce<double>, CustomForward<double>, HostAllreduce<double>);
template std::unique_ptr<Engine<float>> makeCpuEngine<float>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    HostAllreduce<float>, CustomForward<float>, HostAllreduce<double>);

// Instantiate the LM driver here as well.
template LMReport runLM<double>(Engine<double>&, const AlgoOptionLM&,
                                const SolverOptionPCG&);
template LMReport runLM<float>(Engine<float>&, const AlgoOptionLM&,
                               const SolverOptionPCG&);

}  // namespace megba
