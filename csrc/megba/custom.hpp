// Runtime user-defined edge support: the engine gathers per-observation
// parameter leaves as JPV JetVectors, hands them to a user forward()
// callback (typically Python driving the JetVector op layer), and repacks
// the returned residual JetVectors' dual parts into its Jacobian buffers.
// This is the EdgeVector/BaseEdge::forward capability of the reference
// (/root/reference/include/edge/base_edge.h:26-163) without recompilation;
// the built-in BAL edge bypasses it via the fused register-autodiff kernel.
#pragma once

#include <functional>
#include <vector>

#include "jv/jetvector.hpp"

namespace megba {

// (camLeaves[9], ptLeaves[3], meas[2]) -> res[2] (dense, N=12, nItem local).
template <typename T>
using CustomForward = std::function<void(
    const std::vector<JetVec<T>>& camLeaves,
    const std::vector<JetVec<T>>& ptLeaves,
    const std::vector<JetVec<T>>& meas, std::vector<JetVec<T>>& res)>;

}  // namespace megba
