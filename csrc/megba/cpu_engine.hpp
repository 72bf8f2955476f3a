// CPU engine: full LM + Schur-complement PCG on the host (OpenMP).
// Serves as (a) the BASELINE config-1 "CPU/Eigen reference path" equivalent,
// (b) the numerics oracle every HIP kernel is tested against, and (c) the
// world_size>1 correctness testbed (the allreduce hook is pluggable, so
// multi-process CPU tests over gloo exercise the same distributed math the
// RCCL path uses).
#pragma once

#include <functional>
#include <memory>

#include "common.hpp"
#include "custom.hpp"
#include "engine.hpp"
#include "problem.hpp"

namespace megba {

// In-place allreduce over all ranks (no-op when null / worldSize==1).
// op: 's' = sum, 'm' = max.
template <typename T>
using HostAllreduce = std::function<void(T*, std::size_t, char)>;

// `allreduceScalar`: double-typed variant of the same callback used for the
// control-flow scalars (chi2, norms, rho denominator) so fp32 multi-rank
// runs keep full double precision in the reductions that steer LM.
template <typename T>
std::unique_ptr<Engine<T>> makeCpuEngine(
    const BAProblemHost& prob, const ProblemIndex& ix, const ProblemOption& opt,
    HostAllreduce<T> allreduce, CustomForward<T> customForward = nullptr,
    HostAllreduce<double> allreduceScalar = nullptr);

}  // namespace megba
