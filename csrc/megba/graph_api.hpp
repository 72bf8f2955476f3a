// g2o-style incremental graph-construction API (C++).
//
// The reference's native user surface is a vertex/edge graph assembled one
// element at a time and solved in place:
//   BaseVertex/CameraVertex/PointVertex
//     (/root/reference/include/vertex/base_vertex.h:27-230),
//   BaseEdge::appendVertex / setMeasurement / setInformation
//     (/root/reference/include/edge/base_edge.h:26-163),
//   BaseProblem::appendVertex / appendEdge / solve + writeBack
//     (/root/reference/include/problem/base_problem.h:22-83,
//      src/problem/base_problem.cpp:250-272),
// used as in examples/BAL_Double.cpp:60-164.  This header offers the same
// construction style over the array-based core (BAProblemHost + buildIndex
// + Engine + runLM): append, solve, then read each vertex's estimation.
// Eigen-free by design: estimations are plain std::array blocks.
#pragma once

#include <array>
#include <memory>
#include <vector>

#include "common.hpp"
#include "cpu_engine.hpp"
#include "custom.hpp"
#include "lm.hpp"
#include "problem.hpp"
#ifdef MEGBA_WITH_GPU
#include "gpu/gpu_engine.hpp"
#endif

namespace megba {

enum class VertexKind { CAMERA, POINT };

struct BaseVertex {
  VertexKind kind;
  std::vector<double> estimation;  // camDim (camera) or 3 (point)
  bool fixed = false;
  int slot = -1;  // assigned by GraphProblem::appendVertex

  BaseVertex(VertexKind k, const double* est) : kind(k) {
    estimation.assign(est, est + (k == VertexKind::CAMERA ? 9 : 3));
  }
  // Generic-dims vertex (camDim in {9,6,4}; points are 3-dimensional).
  BaseVertex(VertexKind k, const double* est, int dim) : kind(k) {
    MEGBA_CHECK(k == VertexKind::POINT ? dim == 3
                                       : (dim == 9 || dim == 6 || dim == 4),
                "vertex dim outside the compiled set");
    estimation.assign(est, est + dim);
  }
};

struct CameraVertex : BaseVertex {
  explicit CameraVertex(const std::array<double, 9>& est)
      : BaseVertex(VertexKind::CAMERA, est.data()) {}
};

struct PointVertex : BaseVertex {
  explicit PointVertex(const std::array<double, 3>& est)
      : BaseVertex(VertexKind::POINT, est.data()) {}
};

// One observation connecting exactly one camera and one point (the only
// edge kind the reference implements, base_edge.cpp:27-36).
struct ReprojectionEdge {
  BaseVertex* cam = nullptr;
  BaseVertex* pt = nullptr;
  std::array<double, 2> measurement{{0, 0}};
  std::array<double, 3> information{{1, 0, 1}};  // (i00, i01, i11)
  bool hasInformation = false;

  ReprojectionEdge& appendVertex(BaseVertex* v) {
    MEGBA_CHECK(v != nullptr, "appendVertex(nullptr)");
    if (v->kind == VertexKind::CAMERA) {
      MEGBA_CHECK(cam == nullptr, "edge already has a camera vertex");
      cam = v;
    } else {
      MEGBA_CHECK(pt == nullptr, "edge already has a point vertex");
      pt = v;
    }
    return *this;
  }
  ReprojectionEdge& setMeasurement(double u, double v) {
    measurement = {u, v};
    return *this;
  }
  ReprojectionEdge& setInformation(double i00, double i01, double i11) {
    information = {i00, i01, i11};
    hasInformation = true;
    return *this;
  }
};

class GraphProblem {
 public:
  // Vertices are owned by the caller and must outlive the problem; the
  // solved estimations are written back into them (reference writeBack
  // semantics, base_problem.cpp:250-272).
  void appendVertex(BaseVertex* v) {
    MEGBA_CHECK(v != nullptr, "appendVertex(nullptr)");
    MEGBA_CHECK(v->slot < 0, "vertex already appended");
    if (v->kind == VertexKind::CAMERA) {
      v->slot = (int)cams_.size();
      cams_.push_back(v);
    } else {
      v->slot = (int)pts_.size();
      pts_.push_back(v);
    }
  }
  void appendEdge(const ReprojectionEdge& e) {
    MEGBA_CHECK(e.cam && e.pt,
                "edge must connect one CameraVertex and one PointVertex");
    if (e.cam->slot < 0) appendVertex(e.cam);
    if (e.pt->slot < 0) appendVertex(e.pt);
    edges_.push_back(e);
  }
  int64_t numEdges() const { return (int64_t)edges_.size(); }
  int numVertices() const { return (int)(cams_.size() + pts_.size()); }

  // Runtime user-defined residual (C++ counterpart of the reference's
  // BaseEdge::forward, examples/BAL_Double.cpp:16-34): called once per
  // forward pass with per-observation leaf JetVectors; must return resDim
  // dense residual JetVectors.  See examples/bal_custom_edge_cpp.cpp.
  void setCustomForward(CustomForward<double> f) {
    customForward_ = std::move(f);
  }

  // Assemble, run LM on the chosen device, write estimations back.
  LMReport solve(const ProblemOption& popt, const AlgoOptionLM& algo,
                 const SolverOptionPCG& sopt) {
    BAProblemHost prob = assemble();
    ProblemOption opt = popt;
    opt.camDim = prob.camDim;
    opt.ptDim = prob.ptDim;
    opt.resDim = prob.resDim;
    ProblemIndex ix = buildIndex(prob, opt.worldSize);
    std::unique_ptr<Engine<double>> eng;
    if (opt.device == Device::CPU) {
      eng = makeCpuEngine<double>(prob, ix, opt, nullptr, customForward_);
    } else {
#ifdef MEGBA_WITH_GPU
      eng = makeGpuEngine<double>(prob, ix, opt, std::string(),
                                  customForward_);
#else
      MEGBA_CHECK(false, "built without GPU support");
#endif
    }
    LMReport rep = runLM<double>(*eng, algo, sopt);
    const int cd = prob.camDim;
    std::vector<double> camOut(cams_.size() * cd), ptOut(pts_.size() * 3);
    eng->getParams(camOut.data(), ptOut.data());
    for (size_t i = 0; i < cams_.size(); ++i)
      for (int k = 0; k < cd; ++k)
        cams_[i]->estimation[k] = camOut[cd * i + k];
    for (size_t i = 0; i < pts_.size(); ++i)
      for (int k = 0; k < 3; ++k) pts_[i]->estimation[k] = ptOut[3 * i + k];
    return rep;
  }

 private:
  BAProblemHost assemble() const {
    MEGBA_CHECK(!edges_.empty(), "no edges");
    BAProblemHost p;
    p.ncam = (int)cams_.size();
    p.npt = (int)pts_.size();
    p.nobs = (int64_t)edges_.size();
    p.camDim = (int)cams_[0]->estimation.size();
    for (const auto* v : cams_)
      MEGBA_CHECK((int)v->estimation.size() == p.camDim,
                  "mixed camera dims in one problem");
    p.cams.resize((size_t)p.ncam * p.camDim);
    p.pts.resize((size_t)p.npt * 3);
    for (int i = 0; i < p.ncam; ++i)
      for (int k = 0; k < p.camDim; ++k)
        p.cams[(size_t)p.camDim * i + k] = cams_[i]->estimation[k];
    for (int i = 0; i < p.npt; ++i)
      for (int k = 0; k < 3; ++k) p.pts[3 * i + k] = pts_[i]->estimation[k];
    p.camIdx.resize(p.nobs);
    p.ptIdx.resize(p.nobs);
    p.meas.resize(p.nobs * 2);
    bool anyInfo = false, anyFixed = false;
    for (const auto& e : edges_) anyInfo |= e.hasInformation;
    for (const auto* v : cams_) anyFixed |= v->fixed;
    for (const auto* v : pts_) anyFixed |= v->fixed;
    if (anyInfo) p.info.resize(p.nobs * 3);
    for (int64_t k = 0; k < p.nobs; ++k) {
      const auto& e = edges_[k];
      p.camIdx[k] = e.cam->slot;
      p.ptIdx[k] = e.pt->slot;
      p.meas[2 * k] = e.measurement[0];
      p.meas[2 * k + 1] = e.measurement[1];
      if (anyInfo)
        for (int j = 0; j < 3; ++j) p.info[3 * k + j] = e.information[j];
    }
    if (anyFixed) {
      p.camFixed.resize(p.ncam);
      p.ptFixed.resize(p.npt);
      for (int i = 0; i < p.ncam; ++i) p.camFixed[i] = cams_[i]->fixed;
      for (int i = 0; i < p.npt; ++i) p.ptFixed[i] = pts_[i]->fixed;
    }
    return p;
  }

  std::vector<BaseVertex*> cams_, pts_;
  std::vector<ReprojectionEdge> edges_;
  CustomForward<double> customForward_;
};

}  // namespace megba
