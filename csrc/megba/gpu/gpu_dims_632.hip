// Explicit GpuEngine instantiations for (CD,PD,RD) = (6,3,2).
// Split into its own TU so the dimension set compiles in parallel.
#include "gpu_engine_impl.hpp"

namespace megba {
template std::unique_ptr<Engine<double>> makeGpuEngineDims<double, 6, 3, 2>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    const std::string&, CustomForward<double>, HostAllreduce<double>,
    HostAllreduce<double>);
template std::unique_ptr<Engine<float>> makeGpuEngineDims<float, 6, 3, 2>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    const std::string&, CustomForward<float>, HostAllreduce<float>,
    HostAllreduce<double>);
}  // namespace megba
