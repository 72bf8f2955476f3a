// Python bindings for megba_amd.
// The Python layer supplies the raw problem (numpy arrays) and, for
// multi-process CPU runs, an in-place allreduce callback (torch.distributed
// gloo); the GPU path uses RCCL natively (gpu/gpu_engine.hip).
#include <pybind11/functional.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <memory>

#include "megba/bal_functor.hpp"
#include "megba/common.hpp"
#include "megba/custom.hpp"
#include "megba/jv/jetvector.hpp"
#include "megba/cpu_engine.hpp"
#include "megba/engine.hpp"
#include "megba/lm.hpp"
#include "megba/problem.hpp"

#ifdef MEGBA_WITH_GPU
#include "megba/gpu/gpu_engine.hpp"
#endif

namespace py = pybind11;
using namespace megba;

namespace {

// Python-facing JetVector (fp64).
struct PyJetVec {
  JetVec<double> v;
};

template <typename T>
HostAllreduce<T> wrapAllreduce(py::object fn) {
  if (fn.is_none()) return nullptr;
  // Keep the callable alive in the closure; reacquire the GIL per call.
  auto holder = std::make_shared<py::object>(std::move(fn));
  return [holder](T* data, std::size_t n, char op) {
    py::gil_scoped_acquire gil;
    py::array arr(py::dtype::of<T>(), {(py::ssize_t)n}, {(py::ssize_t)sizeof(T)},
                  data, py::none());
    (*holder)(arr, op == 'm' ? "max" : "sum");
  };
}

struct PyProblem {
  BAProblemHost prob;
  ProblemIndex ix;
  ProblemOption opt;
  std::unique_ptr<Engine<double>> engD;
  std::unique_ptr<Engine<float>> engF;
  bool isDouble = true;

  PyProblem(py::array_t<double, py::array::c_style | py::array::forcecast> cams,
            py::array_t<double, py::array::c_style | py::array::forcecast> pts,
            py::array_t<int, py::array::c_style | py::array::forcecast> camIdx,
            py::array_t<int, py::array::c_style | py::array::forcecast> ptIdx,
            py::array_t<double, py::array::c_style | py::array::forcecast> meas,
            py::object info, py::object camFixed, py::object ptFixed) {
    // Block dims are inferred from the array shapes (reference analogue:
    // ProblemOption.N, include/common.h:27-46).  Compiled set:
    // camDim in {9,6,4}, ptDim = 3, resDim in {2,3}.
    MEGBA_CHECK(cams.ndim() == 2 && (cams.shape(1) == 9 || cams.shape(1) == 6 ||
                                     cams.shape(1) == 4),
                "cams must be (ncam, 9|6|4)");
    MEGBA_CHECK(pts.ndim() == 2 && pts.shape(1) == 3, "pts must be (npt,3)");
    MEGBA_CHECK(meas.ndim() == 2 && (meas.shape(1) == 2 || meas.shape(1) == 3),
                "meas must be (nobs, 2|3)");
    prob.camDim = (int)cams.shape(1);
    prob.ptDim = (int)pts.shape(1);
    prob.resDim = (int)meas.shape(1);
    prob.ncam = (int)cams.shape(0);
    prob.npt = (int)pts.shape(0);
    prob.nobs = (int64_t)meas.shape(0);
    prob.cams.assign(cams.data(), cams.data() + cams.size());
    prob.pts.assign(pts.data(), pts.data() + pts.size());
    prob.camIdx.assign(camIdx.data(), camIdx.data() + camIdx.size());
    prob.ptIdx.assign(ptIdx.data(), ptIdx.data() + ptIdx.size());
    prob.meas.assign(meas.data(), meas.data() + meas.size());
    if (!info.is_none()) {
      const int rw = prob.resDim * (prob.resDim + 1) / 2;
      auto infoArr = py::cast<
          py::array_t<double, py::array::c_style | py::array::forcecast>>(info);
      MEGBA_CHECK(infoArr.ndim() == 2 && infoArr.shape(1) == rw &&
                      infoArr.shape(0) == prob.nobs,
                  "info must be (nobs, resDim*(resDim+1)/2) packed-upper "
                  "symmetric (RD=2: w00,w01,w11)");
      prob.info.assign(infoArr.data(), infoArr.data() + infoArr.size());
    }
    auto readMask = [](py::object o, int n, std::vector<uint8_t>& dst) {
      if (o.is_none()) return;
      auto arr = py::cast<
          py::array_t<uint8_t, py::array::c_style | py::array::forcecast>>(o);
      MEGBA_CHECK((int)arr.size() == n, "fixed mask size mismatch");
      dst.assign(arr.data(), arr.data() + arr.size());
    };
    readMask(camFixed, prob.ncam, prob.camFixed);
    readMask(ptFixed, prob.npt, prob.ptFixed);
  }

  void build(const std::string& device, const std::string& dtype, int rank,
             int worldSize, int deviceIndex, const std::string& diff,
             const std::string& schur, const std::string& loss,
             double lossDelta, py::object allreduce, py::object rcclId,
             py::object customForward, py::object intrinsics) {
    opt.rank = rank;
    opt.worldSize = worldSize;
    opt.deviceIndex = deviceIndex;
    opt.camDim = prob.camDim;
    opt.ptDim = prob.ptDim;
    opt.resDim = prob.resDim;
    if (!intrinsics.is_none()) {
      auto arr = py::cast<
          py::array_t<double, py::array::c_style | py::array::forcecast>>(
          intrinsics);
      MEGBA_CHECK(arr.size() == 3, "intrinsics must be [f, k1, k2]");
      for (int i = 0; i < 3; ++i) opt.intr[i] = arr.data()[i];
    }
    MEGBA_CHECK(!customForward.is_none() ||
                    hasBuiltinResidual(prob.camDim, prob.ptDim, prob.resDim),
                "no built-in residual for (camDim,ptDim,resDim) = (" +
                    std::to_string(prob.camDim) + "," +
                    std::to_string(prob.ptDim) + "," +
                    std::to_string(prob.resDim) +
                    "): provide custom_forward (built-ins: (9,3,2) BAL, "
                    "(6,3,2) BAL fixed-intrinsics, (6,3,3) SE3 point)");
    MEGBA_CHECK(diff != "analytical" ||
                    (prob.camDim == 9 && prob.resDim == 2),
                "analytical diff is only available for the BAL (9,3,2) "
                "model");
    MEGBA_CHECK(device == "cpu" || device == "gpu", "device must be cpu|gpu");
    MEGBA_CHECK(dtype == "float64" || dtype == "float32",
                "dtype must be float64|float32");
    MEGBA_CHECK(diff == "auto" || diff == "analytical",
                "diff must be auto|analytical");
    MEGBA_CHECK(schur == "explicit" || schur == "implicit",
                "schur must be explicit|implicit");
    opt.device = device == "gpu" ? Device::GPU : Device::CPU;
    opt.diff = diff == "analytical" ? DiffMode::ANALYTICAL : DiffMode::AUTO;
    opt.schur = schur == "implicit" ? SchurMode::IMPLICIT : SchurMode::EXPLICIT;
    opt.loss = loss == "huber" ? LossKind::HUBER
               : loss == "cauchy" ? LossKind::CAUCHY
                                  : LossKind::NONE;
    MEGBA_CHECK(loss == "none" || loss == "huber" || loss == "cauchy",
                "loss must be none|huber|cauchy");
    opt.lossDelta = lossDelta;
    isDouble = dtype != "float32";
    MEGBA_CHECK(customForward.is_none() || isDouble,
                "custom_forward requires float64");
    CustomForward<double> cf = wrapCustomForward(customForward);
    ix = buildIndex(prob, worldSize);
    if (opt.device == Device::CPU) {
      // A CPU world>1 run without an allreduce hook would silently skip
      // every reduction (each collective is guarded by `if (ar_)`) and
      // produce wrong Hpp/g/chi2 per rank; refuse it like the GPU path
      // refuses a missing rccl_id.
      MEGBA_CHECK(worldSize == 1 || !allreduce.is_none(),
                  "CPU world_size>1 requires an allreduce callback");
      if (isDouble)
        engD = makeCpuEngine<double>(prob, ix, opt,
                                     wrapAllreduce<double>(allreduce), cf,
                                     wrapAllreduce<double>(allreduce));
      else
        engF = makeCpuEngine<float>(prob, ix, opt, wrapAllreduce<float>(allreduce),
                                    nullptr, wrapAllreduce<double>(allreduce));
    } else {
#ifdef MEGBA_WITH_GPU
      std::string id;
      if (!rcclId.is_none()) id = py::cast<std::string>(rcclId);
      if (isDouble)
        engD = makeGpuEngine<double>(prob, ix, opt, id, cf,
                                     wrapAllreduce<double>(allreduce),
                                     wrapAllreduce<double>(allreduce));
      else
        engF = makeGpuEngine<float>(prob, ix, opt, id, nullptr,
                                    wrapAllreduce<float>(allreduce),
                                    wrapAllreduce<double>(allreduce));
#else
      MEGBA_CHECK(false, "built without GPU support");
#endif
    }
  }

  static CustomForward<double> wrapCustomForward(py::object fn);

  template <typename F>
  auto withEngine(F&& f) {
    MEGBA_CHECK(engD || engF, "call build() first");
    if (isDouble) return f(*engD);
    return f(*engF);
  }

  py::dict solve(int maxIter, double tau, double eps1, double eps2,
                 int solverMaxIter, double solverTol, double refuseRatio,
                 bool forceIterations, bool verbose) {
    AlgoOptionLM a;
    a.maxIter = maxIter;
    a.initialRegion = tau;
    a.epsilon1 = eps1;
    a.epsilon2 = eps2;
    a.forceIterations = forceIterations;
    a.verbose = verbose;
    SolverOptionPCG s;
    s.maxIter = solverMaxIter;
    s.tol = solverTol;
    s.refuseRatio = refuseRatio;
    LMReport rep;
    {
      py::gil_scoped_release rel;
      if (isDouble)
        rep = runLM<double>(*engD, a, s);
      else
        rep = runLM<float>(*engF, a, s);
    }
    py::dict d;
    d["final_chi2"] = rep.finalChi2;
    d["accepted"] = rep.acceptedSteps;
    d["rejected"] = rep.rejectedSteps;
    d["total_ms"] = rep.totalMs;
    py::list iters;
    for (const auto& it : rep.iters) {
      py::dict e;
      e["iter"] = it.iter;
      e["accepted"] = it.accepted;
      e["chi2"] = it.chi2;
      e["elapsed_ms"] = it.elapsedMs;
      e["pcg_iters"] = it.pcgIters;
      iters.append(e);
    }
    d["iters"] = iters;
    return d;
  }

  // Step-level LM control (bench.py times exact step counts with this).
  std::unique_ptr<LMSession<double>> sessD;
  std::unique_ptr<LMSession<float>> sessF;
  double lmInit(int maxIter, double tau, double eps1, double eps2,
                int solverMaxIter, double solverTol, double refuseRatio,
                bool forceIterations, bool verbose) {
    AlgoOptionLM a;
    a.maxIter = maxIter;
    a.initialRegion = tau;
    a.epsilon1 = eps1;
    a.epsilon2 = eps2;
    a.forceIterations = forceIterations;
    a.verbose = verbose;
    SolverOptionPCG so;
    so.maxIter = solverMaxIter;
    so.tol = solverTol;
    so.refuseRatio = refuseRatio;
    py::gil_scoped_release rel;
    if (isDouble) {
      sessD = std::make_unique<LMSession<double>>(*engD, a, so);
      return sessD->init();
    }
    sessF = std::make_unique<LMSession<float>>(*engF, a, so);
    return sessF->init();
  }
  py::dict lmStep() {
    MEGBA_CHECK(sessD || sessF, "lm_init first");
    IterLog log;
    {
      py::gil_scoped_release rel;
      log = isDouble ? sessD->step() : sessF->step();
    }
    py::dict e;
    e["iter"] = log.iter;
    e["accepted"] = log.accepted;
    e["chi2"] = log.chi2;
    e["elapsed_ms"] = log.elapsedMs;
    e["pcg_iters"] = log.pcgIters;
    e["stopped"] = isDouble ? sessD->stopped() : sessF->stopped();
    return e;
  }

  // Fine-grained steps for tests.
  double forward() {
    return withEngine([&](auto& e) { return e.forward(); });
  }
  void acceptForward() {
    withEngine([&](auto& e) { e.acceptForward(); return 0; });
  }
  void buildLinearSystem() {
    py::gil_scoped_release rel;
    withEngine([&](auto& e) { e.buildLinearSystem(); return 0; });
  }
  void processDiag(double region) {
    withEngine([&](auto& e) { e.processDiag(region); return 0; });
  }
  int solveLinear(int maxIter, double tol, double refuseRatio) {
    SolverOptionPCG s;
    s.maxIter = maxIter;
    s.tol = tol;
    s.refuseRatio = refuseRatio;
    py::gil_scoped_release rel;
    return withEngine([&](auto& e) { return e.solveLinear(s); });
  }
  void updateParams() {
    withEngine([&](auto& e) { e.updateParams(); return 0; });
  }
  double rhoDenominator(double chi2) {
    return withEngine([&](auto& e) { return e.rhoDenominator(chi2); });
  }
  double deltaXL2() { return withEngine([&](auto& e) { return e.deltaXL2(); }); }
  double xL2() { return withEngine([&](auto& e) { return e.xL2(); }); }
  double gInf() { return withEngine([&](auto& e) { return e.gInf(); }); }

  py::tuple getParams() {
    py::array_t<double> cams({prob.ncam, prob.camDim});
    py::array_t<double> pts({prob.npt, prob.ptDim});
    withEngine([&](auto& e) {
      e.getParams(cams.mutable_data(), pts.mutable_data());
      return 0;
    });
    return py::make_tuple(cams, pts);
  }

  py::dict dump() {
    DenseDump dd = withEngine([&](auto& e) { return e.dump(); });
    py::dict d;
    auto toArr = [](std::vector<double>& v) {
      return py::array_t<double>((py::ssize_t)v.size(), v.data());
    };
    d["e0"] = dd.e0;
    d["e1"] = dd.e1;
    d["r"] = toArr(dd.r);
    d["Jc"] = toArr(dd.Jc);
    d["Jp"] = toArr(dd.Jp);
    d["Hpp"] = toArr(dd.Hpp);
    d["Hll"] = toArr(dd.Hll);
    d["Hpl"] = toArr(dd.Hpl);
    d["g"] = toArr(dd.g);
    d["deltaX"] = toArr(dd.deltaX);
    return d;
  }

  py::dict indexInfo() {
    py::dict d;
    d["ncam"] = ix.ncam;
    d["npt"] = ix.npt;
    d["nobs"] = ix.nobs;
    d["cam_of"] = py::array_t<int>((py::ssize_t)ix.camOf.size(), ix.camOf.data());
    d["pt_of"] = py::array_t<int>((py::ssize_t)ix.ptOf.size(), ix.ptOf.data());
    d["pt_rowptr"] =
        py::array_t<int64_t>((py::ssize_t)ix.ptRowPtr.size(), ix.ptRowPtr.data());
    d["pt_split"] = py::array_t<int>((py::ssize_t)ix.ptSplit.size(), ix.ptSplit.data());
    d["split"] = py::array_t<int64_t>((py::ssize_t)ix.split.size(), ix.split.data());
    d["perm"] = py::array_t<int64_t>((py::ssize_t)ix.perm.size(), ix.perm.data());
    return d;
  }
};

}  // namespace

namespace {

PyJetVec mkJv(py::array_t<double, py::array::c_style | py::array::forcecast> value,
              py::object grad, int N, int grad_pos, bool gpu) {
  const int64_t n = value.size();
  const double* gptr = nullptr;
  py::array_t<double, py::array::c_style | py::array::forcecast> garr;
  if (!grad.is_none()) {
    garr = py::cast<py::array_t<double, py::array::c_style | py::array::forcecast>>(grad);
    MEGBA_CHECK(garr.size() == (py::ssize_t)(N * n), "grad must be (N, nItem)");
    gptr = garr.data();
  }
  return PyJetVec{jvFromHost<double>(value.data(), gptr, n, N, grad_pos, gpu)};
}

py::tuple jvDownload(const PyJetVec& a) {
  py::array_t<double> value((py::ssize_t)a.v.nItem);
  py::array_t<double> grad({(py::ssize_t)a.v.N, (py::ssize_t)a.v.nItem});
  jvToHost<double>(a.v, value.mutable_data(), grad.mutable_data());
  return py::make_tuple(value, grad);
}

std::vector<JetVec<double>> jvList(py::sequence seq) {
  std::vector<JetVec<double>> out;
  for (auto h : seq) out.push_back(py::cast<PyJetVec&>(h).v);
  return out;
}

py::list jvWrap(const std::vector<JetVec<double>>& vs) {
  py::list out;
  for (const auto& v : vs) out.append(PyJetVec{v});
  return out;
}

CustomForward<double> PyProblem::wrapCustomForward(py::object fn) {
  if (fn.is_none()) return nullptr;
  auto holder = std::make_shared<py::object>(std::move(fn));
  return [holder](const std::vector<JetVec<double>>& camL,
                  const std::vector<JetVec<double>>& ptL,
                  const std::vector<JetVec<double>>& meas,
                  std::vector<JetVec<double>>& res) {
    py::gil_scoped_acquire gil;
    py::list c, p, m;
    for (const auto& v : camL) c.append(PyJetVec{v});
    for (const auto& v : ptL) p.append(PyJetVec{v});
    for (const auto& v : meas) m.append(PyJetVec{v});
    py::object out = (*holder)(c, p, m);
    py::sequence seq = py::cast<py::sequence>(out);
    res.clear();
    for (auto h : seq) res.push_back(py::cast<PyJetVec&>(h).v);
  };
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "megba_amd core (MI355X-native distributed bundle adjustment)";

  py::class_<PyJetVec>(m, "JetVector")
      .def(py::init(&mkJv), py::arg("value"), py::arg("grad") = py::none(),
           py::arg("N") = 0, py::arg("grad_pos") = -1, py::arg("gpu") = false)
      .def_property_readonly("n_item", [](const PyJetVec& a) { return a.v.nItem; })
      .def_property_readonly("N", [](const PyJetVec& a) { return a.v.N; })
      .def("to_numpy", &jvDownload);
  m.def("jv_scalar", [](double s, int N) { return PyJetVec{jvScalar<double>(s, N)}; });
  m.def("jv_pool_trim", []() { jvPoolTrim(); });
  m.def("jv_add", [](const PyJetVec& a, const PyJetVec& b) { return PyJetVec{jvBinary(JvOp::Add, a.v, b.v)}; });
  m.def("jv_sub", [](const PyJetVec& a, const PyJetVec& b) { return PyJetVec{jvBinary(JvOp::Sub, a.v, b.v)}; });
  m.def("jv_mul", [](const PyJetVec& a, const PyJetVec& b) { return PyJetVec{jvBinary(JvOp::Mul, a.v, b.v)}; });
  m.def("jv_div", [](const PyJetVec& a, const PyJetVec& b) { return PyJetVec{jvBinary(JvOp::Div, a.v, b.v)}; });
  m.def("jv_neg", [](const PyJetVec& a) { return PyJetVec{jvUnary(JvUnary::Neg, a.v)}; });
  m.def("jv_abs", [](const PyJetVec& a) { return PyJetVec{jvUnary(JvUnary::Abs, a.v)}; });
  m.def("jv_sin", [](const PyJetVec& a) { return PyJetVec{jvUnary(JvUnary::Sin, a.v)}; });
  m.def("jv_cos", [](const PyJetVec& a) { return PyJetVec{jvUnary(JvUnary::Cos, a.v)}; });
  m.def("jv_sqrt", [](const PyJetVec& a) { return PyJetVec{jvUnary(JvUnary::Sqrt, a.v)}; });
  m.def("jv_angle_axis_to_rotation", [](py::sequence aa) { return jvWrap(jvAngleAxisToRotation(jvList(aa))); });
  m.def("jv_normalize_angle", [](const PyJetVec& t) { return PyJetVec{jvNormalizeAngle(t.v)}; });
  m.def("jv_rotation2d", [](const PyJetVec& t) { return jvWrap(jvRotation2D(t.v)); });
  m.def("jv_quaternion_to_rotation", [](py::sequence q) { return jvWrap(jvQuaternionToRotation(jvList(q))); });
  m.def("jv_rotation_to_quaternion", [](py::sequence R) { return jvWrap(jvRotationToQuaternion(jvList(R))); });
  m.def("jv_normalize_quaternion", [](py::sequence q) { return jvWrap(jvNormalizeQuaternion(jvList(q))); });
  m.def("jv_radial_distortion", [](py::sequence p, py::sequence intr) { return PyJetVec{jvRadialDistortion(jvList(p), jvList(intr))}; });

  py::class_<PyProblem>(m, "Problem")
      .def(py::init<py::array_t<double, py::array::c_style | py::array::forcecast>,
                    py::array_t<double, py::array::c_style | py::array::forcecast>,
                    py::array_t<int, py::array::c_style | py::array::forcecast>,
                    py::array_t<int, py::array::c_style | py::array::forcecast>,
                    py::array_t<double, py::array::c_style | py::array::forcecast>,
                    py::object, py::object, py::object>(),
           py::arg("cams"), py::arg("pts"), py::arg("cam_idx"),
           py::arg("pt_idx"), py::arg("meas"), py::arg("info") = py::none(),
           py::arg("cam_fixed") = py::none(), py::arg("pt_fixed") = py::none())
      .def("build", &PyProblem::build, py::arg("device") = "cpu",
           py::arg("dtype") = "float64", py::arg("rank") = 0,
           py::arg("world_size") = 1, py::arg("device_index") = 0,
           py::arg("diff") = "auto", py::arg("schur") = "explicit",
           py::arg("loss") = "none", py::arg("loss_delta") = 1.0,
           py::arg("allreduce") = py::none(), py::arg("rccl_id") = py::none(),
           py::arg("custom_forward") = py::none(),
           py::arg("intrinsics") = py::none())
      .def("solve", &PyProblem::solve, py::arg("max_iter") = 20,
           py::arg("tau") = 1e4, py::arg("epsilon1") = 1.0,
           py::arg("epsilon2") = 1e-10, py::arg("solver_max_iter") = 100,
           py::arg("solver_tol") = 1e-1, py::arg("solver_refuse_ratio") = 1.0,
           py::arg("force_iterations") = false, py::arg("verbose") = true)
      .def("lm_init", &PyProblem::lmInit, py::arg("max_iter") = 1000000,
           py::arg("tau") = 1e4, py::arg("epsilon1") = 1.0,
           py::arg("epsilon2") = 1e-10, py::arg("solver_max_iter") = 100,
           py::arg("solver_tol") = 1e-1, py::arg("solver_refuse_ratio") = 1.0,
           py::arg("force_iterations") = false, py::arg("verbose") = false)
      .def("lm_step", &PyProblem::lmStep)
      .def("forward", &PyProblem::forward)
      .def("accept_forward", &PyProblem::acceptForward)
      .def("build_linear_system", &PyProblem::buildLinearSystem)
      .def("process_diag", &PyProblem::processDiag)
      .def("solve_linear", &PyProblem::solveLinear, py::arg("max_iter") = 100,
           py::arg("tol") = 1e-1, py::arg("refuse_ratio") = 1.0)
      .def("update_params", &PyProblem::updateParams)
      .def("rho_denominator", &PyProblem::rhoDenominator)
      .def("delta_x_l2", &PyProblem::deltaXL2)
      .def("x_l2", &PyProblem::xL2)
      .def("g_inf", &PyProblem::gInf)
      .def("get_params", &PyProblem::getParams)
      .def("dump", &PyProblem::dump)
      .def("index_info", &PyProblem::indexInfo);

#ifdef MEGBA_WITH_GPU
  m.attr("has_gpu_support") = true;
  m.def("rccl_unique_id", []() { return py::bytes(rcclUniqueIdString()); });
  m.def("rccl_preflight",
        [](py::bytes id, int rank, int world, int deviceIndex,
           double timeoutSec) {
          std::string s = id;
          py::gil_scoped_release rel;
          return rcclPreflight(s, rank, world, deviceIndex, timeoutSec);
        },
        py::arg("id"), py::arg("rank"), py::arg("world"),
        py::arg("device_index") = 0, py::arg("timeout_sec") = 120.0);
  m.def("hip_device_count", []() { return hipDeviceCountSafe(); });
  m.def("device_synchronize", []() {
    py::gil_scoped_release rel;
    deviceSynchronize();
  });
  m.def("hip_mem_info", []() {
    auto p = hipMemInfoSafe();
    return py::make_tuple(p.first, p.second);
  });
#else
  m.attr("has_gpu_support") = false;
#endif
}
