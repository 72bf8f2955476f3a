// Engine interface: the device-specific half of the solver.
// CPU (oracle, OpenMP) and GPU (HIP/gfx950) implement the same contract; the
// LM trust-region driver (lm.hpp) is shared host code.
#pragma once

#include <cstdint>
#include <vector>

#include "common.hpp"

namespace megba {

// Everything a test might want to inspect, in double precision.
struct DenseDump {
  int64_t e0 = 0, e1 = 0;  // local edge range (sorted order)
  std::vector<double> r, Jc, Jp;       // accepted set is NOT dumped; these are
                                       // the buffers of the last forward()
  std::vector<double> Hpp, Hll, Hpl;   // [cam][9][9], [pt][3][3], [e][9][3]
  std::vector<double> g, deltaX;       // camera block first
};

template <typename T>
class Engine {
 public:
  virtual ~Engine() = default;

  // Evaluate residuals + Jacobians from the current parameters into the
  // "current" buffer set; returns the global chi^2 (sum over all ranks).
  virtual double forward() = 0;

  // Accept the current forward pass: current r/J become the accepted/backup
  // set used by buildLinearSystem and rhoDenominator (reference:
  // jvBackup = jv).  Must be called after forward(), before
  // buildLinearSystem().
  virtual void acceptForward() = 0;

  // Assemble Hpp/Hll/Hpl/g from the *accepted* buffers, then allreduce
  // Hpp, Hll, g across ranks (reference site A1,
  // /root/reference/src/edge/build_linear_system.cu:403-422).
  virtual void buildLinearSystem() = 0;

  // Parameter state backup / rollback (reference: edges.backup()/rollback()).
  virtual void backupParams() = 0;
  virtual void rollbackParams() = 0;

  // deltaX (and g) backup / rollback (reference: linearSystem.backup()).
  virtual void backupGDx() = 0;
  virtual void rollbackGDx() = 0;

  // Produce damped copies of the Hpp/Hll diagonals: d' = d * (1 + 1/region)
  // (reference processDiag, schur_LM_linear_system.cu:112-185; originals are
  // kept so no recover pass is needed).
  virtual void processDiag(double region) = 0;

  // Distributed Schur-complement PCG; writes deltaX.  Returns #iterations.
  virtual int solveLinear(const SolverOptionPCG& opt) = 0;

  virtual double deltaXL2() = 0;   // ||deltaX||_2 (full vector)
  virtual double xL2() = 0;        // ||x||_2 over current parameters
  virtual double gInf() = 0;       // ||g||_inf

  // params += deltaX (both camera and point blocks; all ranks, replicated).
  virtual void updateParams() = 0;

  // sum over edges of (J*deltaX + r)^2 - chi2Backup, with r/J from the BACKUP
  // set (reference computeRhoDenominator, lm_algo.cu:82-126).
  virtual double rhoDenominator(double chi2Backup) = 0;

  // Introspection (tests / write-back).  COLLECTIVE when worldSize>1: every
  // rank must call it (point shards are merged with an allreduce).
  virtual void getParams(double* cams, double* pts) = 0;
  virtual DenseDump dump() const = 0;
};

}  // namespace megba
