// megba_amd: MI355X-native distributed bundle adjustment.
// Common option structs and enums.
//
// Capability parity targets (reference: /root/reference/include/common.h:17-60):
// ProblemOption / SolverOption.solverOptionPCG / AlgoOption.algoOptionLM carry the
// same knobs with the same semantics (tau == initialRegion, epsilon1/2, PCG
// maxIter/tol/refuseRatio).  The layout here is a plain aggregate threaded by
// value; no global singletons (the reference used static MemoryPool/HandleManager).
#pragma once

#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace megba {

enum class Device { CPU, GPU };
enum class DiffMode { AUTO, ANALYTICAL };
enum class SchurMode { EXPLICIT, IMPLICIT };
// Robust loss (beyond the reference, which only has the 2x2 information
// matrix): IRLS reweighting with rho-consistent cost, s = r^T r.
enum class LossKind { NONE, HUBER, CAUCHY };

struct SolverOptionPCG {
  int maxIter = 100;
  double tol = 1e-1;          // absolute threshold on |r^T z| (reference semantics)
  double refuseRatio = 1.0;   // early stop: rho > refuseRatio * rhoMin -> restore backup
};

struct AlgoOptionLM {
  int maxIter = 20;
  double initialRegion = 1e4;  // "tau": damping is diag * (1 + 1/region)
  double epsilon1 = 1.0;       // gradient inf-norm stop + used in epsilon2 criterion
  double epsilon2 = 1e-10;     // ||dx|| <= eps2 * (||x|| + eps1) stop
  // Benchmark mode: ignore stop criteria and run exactly maxIter iterations.
  bool forceIterations = false;
  bool verbose = true;
};

struct ProblemOption {
  Device device = Device::CPU;
  DiffMode diff = DiffMode::AUTO;
  SchurMode schur = SchurMode::EXPLICIT;
  LossKind loss = LossKind::NONE;
  double lossDelta = 1.0;
  int rank = 0;        // this process' rank (one process per GPU)
  int worldSize = 1;
  int deviceIndex = 0; // HIP device ordinal for this rank
  // Block dimensions (camera/point/residual).  The engines are compiled for
  // a fixed practical set ({9,6,4} x {3} x {2,3}); the reference took these
  // as runtime values everywhere (build_linear_system.cu:48-146,
  // common.h:27-46 ProblemOption.N).  Dims select the built-in residual:
  // (9,3,2) BAL, (6,3,2) BAL with fixed intrinsics, (6,3,3) SE3 point
  // alignment; other combinations require a custom forward.
  int camDim = 9;
  int ptDim = 3;
  int resDim = 2;
  double intr[3] = {1.0, 0.0, 0.0};  // f,k1,k2 for the (6,3,2) built-in
};

struct IterLog {
  int iter = 0;          // LM iteration number (0 = initial state)
  bool accepted = true;
  double chi2 = 0.0;     // sum of squared residuals (error printed as chi2/2)
  double elapsedMs = 0.0;
  int pcgIters = 0;
};

struct LMReport {
  std::vector<IterLog> iters;
  double finalChi2 = 0.0;
  int acceptedSteps = 0;
  int rejectedSteps = 0;
  double totalMs = 0.0;
};

// Fixed BAL-family block dimensions (camera 9 = angle-axis 3 + t 3 + f,k1,k2;
// point 3; residual 2).  The kernels are written for these sizes; the problem
// layer validates them.
constexpr int kCamDim = 9;
constexpr int kPtDim = 3;
constexpr int kResDim = 2;
constexpr int kGradW = kCamDim + kPtDim;  // 12

#define MEGBA_CHECK(cond, msg)                         \
  do {                                                 \
    if (!(cond)) throw std::runtime_error(std::string("megba: ") + (msg)); \
  } while (0)

}  // namespace megba
