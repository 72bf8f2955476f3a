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
#pragma once

#include <chrono>
#include <cmath>
#include <cstdio>

#include "common.hpp"
#include "engine.hpp"

namespace megba {

template <typename T>
LMReport runLM(Engine<T>& eng, const AlgoOptionLM& opt,
               const SolverOptionPCG& sopt) {
  using Clock = std::chrono::steady_clock;
  const auto t0 = Clock::now();
  auto elapsedMs = [&]() {
    return std::chrono::duration<double, std::milli>(Clock::now() - t0).count();
  };

  LMReport rep;
  double chi2New = eng.forward();
  eng.acceptForward();
  eng.buildLinearSystem();
  eng.backupGDx();
  eng.backupParams();
  double chi2 = chi2New;
  if (opt.verbose) {
    std::printf("Start with error: %.10g, log error: %.6f, elapsed %.1f ms\n",
                chi2New / 2, std::log10(chi2New / 2), elapsedMs());
    std::fflush(stdout);
  }
  rep.iters.push_back({0, true, chi2New, elapsedMs(), 0});

  bool stop = false;
  double v = 2.0;
  double region = opt.initialRegion;
  int k = 0;
  while ((!stop || opt.forceIterations) && k < opt.maxIter) {
    ++k;
    eng.processDiag(region);
    const int pcgIters = eng.solveLinear(sopt);
    const double deltaXL2 = eng.deltaXL2();
    const double xL2 = eng.xL2();
    if (!opt.forceIterations &&
        deltaXL2 <= opt.epsilon2 * (xL2 + opt.epsilon1)) {
      break;
    }
    eng.updateParams();
    const double rhoDenominator = eng.rhoDenominator(chi2New);
    const double chi2Old = chi2New;
    chi2New = eng.forward();
    const double rho = -(chi2Old - chi2New) / rhoDenominator;
    IterLog log;
    log.iter = k;
    log.pcgIters = pcgIters;
    if (chi2Old > chi2New) {
      log.accepted = true;
      log.chi2 = chi2New;
      eng.acceptForward();
      eng.buildLinearSystem();
      eng.backupGDx();
      eng.backupParams();
      chi2 = chi2New;
      region /= std::max(1.0 / 3.0, 1.0 - std::pow(2.0 * rho - 1.0, 3.0));
      v = 2.0;
      rep.acceptedSteps++;
      const double gnorm = eng.gInf();
      stop = gnorm <= opt.epsilon1;
      if (opt.verbose) {
        std::printf("Iter %d error: %.10g, log error: %.6f, elapsed %.1f ms\n",
                    k, chi2New / 2, std::log10(chi2New / 2), elapsedMs());
        std::fflush(stdout);
      }
    } else {
      log.accepted = false;
      log.chi2 = chi2Old;
      eng.rollbackParams();
      eng.rollbackGDx();
      chi2New = chi2Old;
      region /= v;
      v *= 2.0;
      rep.rejectedSteps++;
      if (opt.verbose) {
        std::printf("Iter %d failed, elapsed %.1f ms\n", k, elapsedMs());
        std::fflush(stdout);
      }
    }
    log.elapsedMs = elapsedMs();
    rep.iters.push_back(log);
  }
  rep.finalChi2 = chi2;
  rep.totalMs = elapsedMs();
  if (opt.verbose) {
    std::printf("Finished\n");
    std::fflush(stdout);
  }
  return rep;
}

}  // namespace megba
