// Levenberg-Marquardt trust-region driver (device-agnostic host code).
//
// Control flow mirrors the reference LM loop exactly
// (/root/reference/src/algo/lm_algo.cu:139-223):
//   region' = region / max(1/3, 1-(2*rho-1)^3) on accept,
//   region' = region / v; v *= 2 on reject;
//   damping d' = d * (1 + 1/region);
//   stop on ||dx|| <= eps2*(||x||+eps1) or ||g||inf <= eps1.
// The per-iteration log line format ("Iter k error: E, log error: L, elapsed
// M ms") is kept as the de-facto UX (lm_algo.cu:190-220).
//
// LMSession exposes the loop one iteration at a time so bench.py can time an
// exact number of steps; runLM() is the batch driver on top of it.
#pragma once

#include <chrono>
#include <cmath>
#include <cstdio>

#include "common.hpp"
#include "engine.hpp"

namespace megba {

template <typename T>
class LMSession {
 public:
  LMSession(Engine<T>& eng, const AlgoOptionLM& opt,
            const SolverOptionPCG& sopt)
      : eng_(eng), opt_(opt), sopt_(sopt) {}

  // Initial forward + assembly (iteration 0). Returns initial chi2.
  double init() {
    t0_ = Clock::now();
    chi2New_ = eng_.forward();
    eng_.acceptForward();
    eng_.buildLinearSystem();
    eng_.backupGDx();
    eng_.backupParams();
    region_ = opt_.initialRegion;
    v_ = 2.0;
    k_ = 0;
    stop_ = false;
    if (opt_.verbose) {
      std::printf("Start with error: %.10g, log error: %.6f, elapsed %.1f ms\n",
                  chi2New_ / 2, std::log10(chi2New_ / 2), elapsedMs());
      std::fflush(stdout);
    }
    inited_ = true;
    return chi2New_;
  }

  bool stopped() const { return stop_; }
  int iter() const { return k_; }
  double chi2() const { return chi2New_; }

  // One LM iteration. Returns the iteration log entry.
  IterLog step() {
    MEGBA_CHECK(inited_, "LMSession::init() first");
    ++k_;
    IterLog log;
    log.iter = k_;
    eng_.processDiag(region_);
    log.pcgIters = eng_.solveLinear(sopt_);
    const double deltaXL2 = eng_.deltaXL2();
    const double xL2 = eng_.xL2();
    if (!opt_.forceIterations &&
        deltaXL2 <= opt_.epsilon2 * (xL2 + opt_.epsilon1)) {
      stop_ = true;
      log.accepted = false;
      log.chi2 = chi2New_;
      log.elapsedMs = elapsedMs();
      return log;
    }
    eng_.updateParams();
    const double rhoDenominator = eng_.rhoDenominator(chi2New_);
    const double chi2Old = chi2New_;
    chi2New_ = eng_.forward();
    const double rho = -(chi2Old - chi2New_) / rhoDenominator;
    if (chi2Old > chi2New_) {
      log.accepted = true;
      log.chi2 = chi2New_;
      eng_.acceptForward();
      eng_.buildLinearSystem();
      eng_.backupGDx();
      eng_.backupParams();
      region_ /= std::max(1.0 / 3.0, 1.0 - std::pow(2.0 * rho - 1.0, 3.0));
      v_ = 2.0;
      const double gnorm = eng_.gInf();
      if (!opt_.forceIterations && gnorm <= opt_.epsilon1) stop_ = true;
      if (opt_.verbose) {
        std::printf("Iter %d error: %.10g, log error: %.6f, elapsed %.1f ms\n",
                    k_, chi2New_ / 2, std::log10(chi2New_ / 2), elapsedMs());
        std::fflush(stdout);
      }
    } else {
      log.accepted = false;
      log.chi2 = chi2Old;
      eng_.rollbackParams();
      eng_.rollbackGDx();
      chi2New_ = chi2Old;
      region_ /= v_;
      v_ *= 2.0;
      if (opt_.verbose) {
        std::printf("Iter %d failed, elapsed %.1f ms\n", k_, elapsedMs());
        std::fflush(stdout);
      }
    }
    log.elapsedMs = elapsedMs();
    return log;
  }

 private:
  using Clock = std::chrono::steady_clock;
  double elapsedMs() const {
    return std::chrono::duration<double, std::milli>(Clock::now() - t0_).count();
  }
  Engine<T>& eng_;
  AlgoOptionLM opt_;
  SolverOptionPCG sopt_;
  Clock::time_point t0_;
  double chi2New_ = 0, region_ = 0, v_ = 2.0;
  int k_ = 0;
  bool stop_ = false, inited_ = false;
};

template <typename T>
LMReport runLM(Engine<T>& eng, const AlgoOptionLM& opt,
               const SolverOptionPCG& sopt) {
  LMSession<T> s(eng, opt, sopt);
  LMReport rep;
  const double chi0 = s.init();
  rep.iters.push_back({0, true, chi0, 0.0, 0});
  while (!s.stopped() && s.iter() < opt.maxIter) {
    IterLog log = s.step();
    rep.iters.push_back(log);
    if (log.accepted)
      rep.acceptedSteps++;
    else if (!s.stopped())
      rep.rejectedSteps++;
    rep.totalMs = log.elapsedMs;
  }
  rep.finalChi2 = s.chi2();
  if (opt.verbose) {
    std::printf("Finished\n");
    std::fflush(stdout);
  }
  return rep;
}

}  // namespace megba
