// CPU engine: dims dispatch + instantiations.
// The implementation (CpuEngine<T, CD, PD, RD>) lives in cpu_engine_impl.hpp;
// this TU instantiates the practical dimension set and routes runtime dims
// to the right instantiation (the reference's equivalent is runtime
// cameraDim/pointDim/resDim kernel arguments, build_linear_system.cu:48-146).
#include "cpu_engine_impl.hpp"

namespace megba {

template <typename T>
std::unique_ptr<Engine<T>> makeCpuEngine(const BAProblemHost& prob,
                                         const ProblemIndex& ix,
                                         const ProblemOption& opt,
                                         HostAllreduce<T> allreduce,
                                         CustomForward<T> customForward,
                                         HostAllreduce<double> allreduceScalar) {
  const int cd = prob.camDim, pd = prob.ptDim, rd = prob.resDim;
#define MEGBA_CPU_CASE(CDv, PDv, RDv)                                     \
  if (cd == CDv && pd == PDv && rd == RDv)                                \
    return std::make_unique<CpuEngine<T, CDv, PDv, RDv>>(                 \
        prob, ix, opt, std::move(allreduce), std::move(customForward),    \
        std::move(allreduceScalar));
  MEGBA_CPU_CASE(9, 3, 2)
  MEGBA_CPU_CASE(6, 3, 2)
  MEGBA_CPU_CASE(4, 3, 2)
  MEGBA_CPU_CASE(9, 3, 3)
  MEGBA_CPU_CASE(6, 3, 3)
  MEGBA_CPU_CASE(4, 3, 3)
#undef MEGBA_CPU_CASE
  MEGBA_CHECK(false,
              "unsupported (camDim,ptDim,resDim) = (" + std::to_string(cd) +
                  "," + std::to_string(pd) + "," + std::to_string(rd) +
                  "); compiled set: {9,6,4} x {3} x {2,3}");
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
