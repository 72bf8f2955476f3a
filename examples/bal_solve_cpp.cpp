// Native C++ BAL solver CLI — the direct counterpart of the reference's
// BAL_Double binary (examples/BAL_Double.cpp), built on the C++ core with no
// Python involved.  Single-process (world_size 1); the multi-GPU launcher is
// examples/bal_solve.py (torchrun, one rank per GPU).
//
// Build:  python build.py --examples      (emits examples/bal_solve_cpp)
// Run:    examples/bal_solve_cpp --path problem-49-7776-pre.txt \
//             --device gpu --max_iter 20 --tau 1e4
#include <cstring>
#include <fstream>
#include <iostream>
#include <sstream>
#include <string>

#include "megba/common.hpp"
#include "megba/graph_api.hpp"
#include "megba/problem.hpp"

using namespace megba;

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

int main(int argc, char** argv) {
  std::string path, device = "gpu", diff = "auto", schur = "explicit";
  std::string loss = "none";
  double lossDelta = 1.0;
  AlgoOptionLM algo;
  SolverOptionPCG sopt;
  sopt.maxIter = 50;
  sopt.tol = 10.0;
  algo.initialRegion = 1.0;
  for (int i = 1; i < argc; ++i) {
    auto arg = [&](const char* name) {
      return std::strcmp(argv[i], name) == 0 && i + 1 < argc;
    };
    if (arg("--path")) path = argv[++i];
    else if (arg("--device")) device = argv[++i];
    else if (arg("--diff")) diff = argv[++i];
    else if (arg("--schur")) schur = argv[++i];
    else if (arg("--max_iter")) algo.maxIter = std::atoi(argv[++i]);
    else if (arg("--solver_max_iter")) sopt.maxIter = std::atoi(argv[++i]);
    else if (arg("--solver_tol")) sopt.tol = std::atof(argv[++i]);
    else if (arg("--solver_refuse_ratio")) sopt.refuseRatio = std::atof(argv[++i]);
    else if (arg("--tau")) algo.initialRegion = std::atof(argv[++i]);
    else if (arg("--epsilon1")) algo.epsilon1 = std::atof(argv[++i]);
    else if (arg("--epsilon2")) algo.epsilon2 = std::atof(argv[++i]);
    else if (arg("--loss")) loss = argv[++i];
    else if (arg("--loss_delta")) lossDelta = std::atof(argv[++i]);
    else {
      std::cerr << "unknown/incomplete flag " << argv[i] << "\n";
      return 2;
    }
  }
  if (path.empty()) {
    std::cerr << "usage: bal_solve_cpp --path problem.txt [--device gpu|cpu] "
                 "[--diff auto|analytical] [--schur explicit|implicit] "
                 "[--max_iter N] [--solver_* ...] [--tau T]\n";
    return 2;
  }
  BAProblemHost raw = loadBal(path);
  std::cout << "solving " << path << " (" << raw.ncam << " cams, " << raw.npt
            << " pts, " << raw.nobs << " obs), device=" << device << "\n";

  // Build through the g2o-style graph API — the same construction flow as
  // the reference demo (examples/BAL_Double.cpp:60-164: one vertex per
  // camera/point, one edge per observation, then solve + writeBack).
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
    e.appendVertex(&camVs[raw.camIdx[k]]).appendVertex(&ptVs[raw.ptIdx[k]])
        .setMeasurement(raw.meas[2 * k], raw.meas[2 * k + 1]);
    graph.appendEdge(e);
  }

  ProblemOption opt;
  opt.device = device == "cpu" ? Device::CPU : Device::GPU;
  opt.diff = diff == "analytical" ? DiffMode::ANALYTICAL : DiffMode::AUTO;
  opt.schur = schur == "implicit" ? SchurMode::IMPLICIT : SchurMode::EXPLICIT;
  opt.loss = loss == "huber" ? LossKind::HUBER
             : loss == "cauchy" ? LossKind::CAUCHY
                                : LossKind::NONE;
  opt.lossDelta = lossDelta;
  LMReport rep = graph.solve(opt, algo, sopt);
  std::cout << "final error: " << rep.finalChi2 / 2 << " after "
            << rep.acceptedSteps << " accepted / " << rep.rejectedSteps
            << " rejected steps, " << rep.totalMs << " ms\n";
  return 0;
}
