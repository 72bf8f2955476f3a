// Native C++ custom-edge example — the counterpart of the reference's
// user-defined `BaseEdge::forward` (examples/BAL_Double.cpp:16-34): the
// BAL reprojection residual is expressed as a C++ callback over the
// vectorised JetVector op layer (one op = one fused kernel over all local
// observations) instead of the engine's built-in fused kernel, with NO
// recompilation of the library.  The same callback type serves any
// residual and any compiled (camDim, ptDim, resDim).
//
// Runs the SAME problem through the built-in fused path and the custom
// path and checks the LM trajectories agree — a self-verifying example.
//
// Build: python build.py            (emits examples/bal_custom_edge_cpp)
// Run:   examples/bal_custom_edge_cpp --path problem.txt [--device gpu|cpu]
#include <cmath>
#include <cstring>
#include <fstream>
#include <iostream>
#include <string>
#include <vector>

#include "megba/common.hpp"
#include "megba/graph_api.hpp"
#include "megba/jv/jetvector.hpp"
#include "megba/problem.hpp"

using namespace megba;
using JV = JetVec<double>;

// The BAL reprojection residual over JetVectors (all-observation ops).
static void balCustomForward(const std::vector<JV>& cam,
                             const std::vector<JV>& pt,
                             const std::vector<JV>& meas,
                             std::vector<JV>& res) {
  auto add = [](const JV& a, const JV& b) { return jvBinary(JvOp::Add, a, b); };
  auto sub = [](const JV& a, const JV& b) { return jvBinary(JvOp::Sub, a, b); };
  auto mul = [](const JV& a, const JV& b) { return jvBinary(JvOp::Mul, a, b); };
  auto div = [](const JV& a, const JV& b) { return jvBinary(JvOp::Div, a, b); };
  // R = Rodrigues(cam[0:3]) -- 9 JetVectors, row-major
  std::vector<JV> R =
      jvAngleAxisToRotation(std::vector<JV>{cam[0], cam[1], cam[2]});
  JV P[3];
  for (int i = 0; i < 3; ++i) {
    JV s = mul(R[3 * i + 0], pt[0]);
    s = add(s, mul(R[3 * i + 1], pt[1]));
    s = add(s, mul(R[3 * i + 2], pt[2]));
    P[i] = add(s, cam[3 + i]);
  }
  JV negz = jvUnary(JvUnary::Neg, P[2]);
  JV px = div(P[0], negz);
  JV py = div(P[1], negz);
  // f * (1 + k1 r^2 + k2 r^4)
  JV fr = jvRadialDistortion(std::vector<JV>{px, py},
                             std::vector<JV>{cam[6], cam[7], cam[8]});
  res.clear();
  res.push_back(sub(mul(fr, px), meas[0]));
  res.push_back(sub(mul(fr, py), meas[1]));
}

static BAProblemHost loadBal(const std::string& path) {
  std::ifstream f(path);
  MEGBA_CHECK(f.good(), "cannot open " + path);
  BAProblemHost p;
  f >> p.ncam >> p.npt >> p.nobs;
  p.camIdx.resize(p.nobs);
  p.ptIdx.resize(p.nobs);
  p.meas.resize(p.nobs * 2);
  for (int64_t i = 0; i < p.nobs; ++i)
    f >> p.camIdx[i] >> p.ptIdx[i] >> p.meas[2 * i] >> p.meas[2 * i + 1];
  p.cams.resize((size_t)p.ncam * 9);
  for (auto& v : p.cams) f >> v;
  p.pts.resize((size_t)p.npt * 3);
  for (auto& v : p.pts) f >> v;
  MEGBA_CHECK(f.good() || f.eof(), "truncated BAL file");
  return p;
}

static LMReport solveOnce(const BAProblemHost& raw, const std::string& device,
                          bool custom, int maxIter) {
  std::vector<BaseVertex> camVs, ptVs;
  camVs.reserve(raw.ncam);
  ptVs.reserve(raw.npt);
  for (int i = 0; i < raw.ncam; ++i)
    camVs.emplace_back(VertexKind::CAMERA, &raw.cams[9 * i]);
  for (int i = 0; i < raw.npt; ++i)
    ptVs.emplace_back(VertexKind::POINT, &raw.pts[3 * i]);
  GraphProblem graph;
  for (auto& v : camVs) graph.appendVertex(&v);
  for (auto& v : ptVs) graph.appendVertex(&v);
  for (int64_t k = 0; k < raw.nobs; ++k) {
    ReprojectionEdge e;
    e.appendVertex(&camVs[raw.camIdx[k]])
        .appendVertex(&ptVs[raw.ptIdx[k]])
        .setMeasurement(raw.meas[2 * k], raw.meas[2 * k + 1]);
    graph.appendEdge(e);
  }
  if (custom) graph.setCustomForward(balCustomForward);
  ProblemOption opt;
  opt.device = device == "cpu" ? Device::CPU : Device::GPU;
  AlgoOptionLM algo;
  algo.maxIter = maxIter;
  algo.verbose = false;
  SolverOptionPCG sopt;
  sopt.maxIter = 50;
  sopt.tol = 1e-6;
  sopt.refuseRatio = 1e6;
  return graph.solve(opt, algo, sopt);
}

int main(int argc, char** argv) {
  std::string path, device = "gpu";
  int maxIter = 6;
  for (int i = 1; i < argc; ++i) {
    auto arg = [&](const char* name) {
      return std::strcmp(argv[i], name) == 0 && i + 1 < argc;
    };
    if (arg("--path")) path = argv[++i];
    else if (arg("--device")) device = argv[++i];
    else if (arg("--max_iter")) maxIter = std::atoi(argv[++i]);
    else {
      std::cerr << "usage: bal_custom_edge_cpp --path problem.txt "
                   "[--device gpu|cpu] [--max_iter N]\n";
      return 2;
    }
  }
  if (path.empty()) {
    std::cerr << "usage: bal_custom_edge_cpp --path problem.txt\n";
    return 2;
  }
  BAProblemHost raw = loadBal(path);
  std::cout << "problem: " << raw.ncam << " cams, " << raw.npt << " pts, "
            << raw.nobs << " obs; device=" << device << "\n";
  LMReport builtin = solveOnce(raw, device, false, maxIter);
  LMReport custom = solveOnce(raw, device, true, maxIter);
  std::cout << "builtin final error: " << builtin.finalChi2 / 2 << "\n";
  std::cout << "custom  final error: " << custom.finalChi2 / 2 << "\n";
  const double rel = std::fabs(builtin.finalChi2 - custom.finalChi2) /
                     std::max(builtin.finalChi2, 1e-30);
  MEGBA_CHECK(builtin.iters.size() == custom.iters.size(),
              "trajectory length mismatch");
  for (size_t k = 0; k < builtin.iters.size(); ++k) {
    const double r = std::fabs(builtin.iters[k].chi2 - custom.iters[k].chi2) /
                     std::max(builtin.iters[k].chi2, 1e-30);
    MEGBA_CHECK(r < 1e-6, "trajectory diverged at iter " + std::to_string(k));
  }
  std::cout << "CUSTOM_MATCH_OK (rel " << rel << ")\n";
  return 0;
}
