// GPU engine factory + RCCL bootstrap helpers (host-callable API).
// Implementation: gpu_engine.hip (HIP kernels for gfx950 + host orchestration).
#pragma once

#include <memory>
#include <string>

#include "../common.hpp"
#include "../custom.hpp"
#include "../engine.hpp"
#include "../problem.hpp"

namespace megba {

// `rcclId`: empty for worldSize==1; otherwise the ncclUniqueId bytes created
// by rank 0 (rcclUniqueIdString) and broadcast out-of-band (bench.py uses a
// torch.distributed gloo store for the exchange; RCCL itself then runs
// natively over xGMI with no Python in the loop).
template <typename T>
std::unique_ptr<Engine<T>> makeGpuEngine(const BAProblemHost& prob,
                                         const ProblemIndex& ix,
                                         const ProblemOption& opt,
                                         const std::string& rcclId,
                                         CustomForward<T> customForward = nullptr);

std::string rcclUniqueIdString();
int hipDeviceCountSafe();

}  // namespace megba
