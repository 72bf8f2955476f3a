// GPU engine factory + RCCL bootstrap helpers (host-callable API).
// Implementation: gpu_engine.hip (HIP kernels for gfx950 + host orchestration).
#pragma once

#include <memory>
#include <string>
#include <utility>

#include "../common.hpp"
#include "../cpu_engine.hpp"  // HostAllreduce
#include "../custom.hpp"
#include "../engine.hpp"
#include "../problem.hpp"

namespace megba {

// `rcclId`: empty for worldSize==1; otherwise the ncclUniqueId bytes created
// by rank 0 (rcclUniqueIdString) and broadcast out-of-band (bench.py uses a
// torch.distributed gloo store for the exchange; RCCL itself then runs
// natively over xGMI with no Python in the loop).
// `hostAllreduce`: testing fallback — when worldSize>1 and rcclId is empty,
// collectives bounce through the host callback (e.g. torch.distributed gloo),
// so the sharded GPU code paths can be exercised by several ranks sharing
// one GPU.  Production multi-GPU uses RCCL (rcclId non-empty).
// `hostAllreduceScalar`: double-typed variant of the host callback used for
// control-flow scalars so fp32 gloo-fallback runs reduce in full double.
template <typename T>
std::unique_ptr<Engine<T>> makeGpuEngine(
    const BAProblemHost& prob, const ProblemIndex& ix, const ProblemOption& opt,
    const std::string& rcclId, CustomForward<T> customForward = nullptr,
    HostAllreduce<T> hostAllreduce = nullptr,
    HostAllreduce<double> hostAllreduceScalar = nullptr);

std::string rcclUniqueIdString();
// Bootstrap self-test: comm init + 1-element allreduce under a watchdog;
// throws (instead of hanging) when a peer is missing.  Returns seconds.
double rcclPreflight(const std::string& idBytes, int rank, int world,
                     int deviceIndex, double timeoutSec);
int hipDeviceCountSafe();
void deviceSynchronize();  // hipDeviceSynchronize (bench timing fences)
std::pair<long long, long long> hipMemInfoSafe();  // (free, total) bytes

}  // namespace megba
