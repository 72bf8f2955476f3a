// GPU engine: dims dispatch + RCCL/HIP utilities.
// The templated implementation (GpuEngine<T, CD, PD, RD> + all kernels)
// lives in gpu_engine_impl.hpp and is explicitly instantiated by the
// gpu_dims_*.hip TUs (parallel compilation); this TU routes runtime dims
// to the right instantiation (the reference passed cameraDim/pointDim/
// resDim as runtime kernel arguments, build_linear_system.cu:48-146).
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cmath>
#include <condition_variable>
#include <cstring>
#include <memory>
#include <mutex>
#include <string>
#include <thread>

#include "gpu_engine.hpp"

namespace megba {

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    MEGBA_CHECK(_e == hipSuccess,                                           \
                std::string("HIP error: ") + hipGetErrorString(_e) + " @ " #expr); \
  } while (0)

#define RCCL_CHECK(expr)                                                    \
  do {                                                                      \
    ncclResult_t _e = (expr);                                               \
    MEGBA_CHECK(_e == ncclSuccess,                                          \
                std::string("RCCL error: ") + ncclGetErrorString(_e) + " @ " #expr); \
  } while (0)

// Per-dims factories defined in the gpu_dims_*.hip TUs.
template <typename T, int CD, int PD, int RD>
std::unique_ptr<Engine<T>> makeGpuEngineDims(const BAProblemHost& prob,
                                             const ProblemIndex& ix,
                                             const ProblemOption& opt,
                                             const std::string& rcclId,
                                             CustomForward<T> customForward,
                                             HostAllreduce<T> hostAllreduce,
                                             HostAllreduce<double> hostArD);

template <typename T>
std::unique_ptr<Engine<T>> makeGpuEngine(const BAProblemHost& prob,
                                         const ProblemIndex& ix,
                                         const ProblemOption& opt,
                                         const std::string& rcclId,
                                         CustomForward<T> customForward,
                                         HostAllreduce<T> hostAllreduce,
                                         HostAllreduce<double> hostArD) {
  const int cd = prob.camDim, pd = prob.ptDim, rd = prob.resDim;
#define MEGBA_GPU_CASE(CDv, PDv, RDv)                                     \
  if (cd == CDv && pd == PDv && rd == RDv)                                \
    return makeGpuEngineDims<T, CDv, PDv, RDv>(                           \
        prob, ix, opt, rcclId, std::move(customForward),                  \
        std::move(hostAllreduce), std::move(hostArD));
  MEGBA_GPU_CASE(9, 3, 2)
  MEGBA_GPU_CASE(6, 3, 2)
  MEGBA_GPU_CASE(4, 3, 2)
  MEGBA_GPU_CASE(9, 3, 3)
  MEGBA_GPU_CASE(6, 3, 3)
  MEGBA_GPU_CASE(4, 3, 3)
#undef MEGBA_GPU_CASE
  MEGBA_CHECK(false,
              "unsupported (camDim,ptDim,resDim) = (" + std::to_string(cd) +
                  "," + std::to_string(pd) + "," + std::to_string(rd) +
                  "); compiled set: {9,6,4} x {3} x {2,3}");
}

template std::unique_ptr<Engine<double>> makeGpuEngine<double>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    const std::string&, CustomForward<double>, HostAllreduce<double>,
    HostAllreduce<double>);
template std::unique_ptr<Engine<float>> makeGpuEngine<float>(
    const BAProblemHost&, const ProblemIndex&, const ProblemOption&,
    const std::string&, CustomForward<float>, HostAllreduce<float>,
    HostAllreduce<double>);

// Pre-flight bootstrap self-test for the driver's multi-GPU scale run:
// init a throwaway communicator from a dedicated unique id, run a
// 1-element allreduce, verify the sum, tear down — all under a watchdog so
// a wedged rendezvous aborts with a clear message instead of hanging the
// whole job (the unique id is single-use; the engine gets its own).
// Returns elapsed seconds.  Reference anchor for what is being guarded:
// /root/reference/src/resource/handle_manager.cpp:17-21 (comm init) + the
// collective sites of SURVEY.md section 2b.
double rcclPreflight(const std::string& idBytes, int rank, int world,
                     int deviceIndex, double timeoutSec) {
  MEGBA_CHECK(idBytes.size() == sizeof(ncclUniqueId),
              "rccl_preflight: bad unique id size");
  struct State {
    std::mutex m;
    std::condition_variable cv;
    bool done = false;
    std::string err;
  };
  auto st = std::make_shared<State>();
  const auto t0 = std::chrono::steady_clock::now();
  ncclUniqueId id;
  std::memcpy(&id, idBytes.data(), sizeof(id));
  std::thread([st, id, rank, world, deviceIndex]() {
    std::string err;
    try {
      HIP_CHECK(hipSetDevice(deviceIndex));
      ncclComm_t comm{};
      RCCL_CHECK(ncclCommInitRank(&comm, world, id, rank));
      double* buf = nullptr;
      HIP_CHECK(hipMalloc((void**)&buf, sizeof(double)));
      const double one = 1.0;
      HIP_CHECK(hipMemcpy(buf, &one, sizeof(double), hipMemcpyHostToDevice));
      RCCL_CHECK(ncclAllReduce(buf, buf, 1, ncclDouble, ncclSum, comm,
                               /*stream=*/0));
      HIP_CHECK(hipStreamSynchronize(0));
      double out = 0.0;
      HIP_CHECK(hipMemcpy(&out, buf, sizeof(double), hipMemcpyDeviceToHost));
      (void)hipFree(buf);
      (void)ncclCommDestroy(comm);
      MEGBA_CHECK(std::abs(out - (double)world) < 1e-9,
                  "rccl_preflight: allreduce sum mismatch");
    } catch (const std::exception& e) {
      err = e.what();
    }
    {
      std::lock_guard<std::mutex> lk(st->m);
      st->done = true;
      st->err = err;
    }
    st->cv.notify_all();
  }).detach();
  std::unique_lock<std::mutex> lk(st->m);
  if (!st->cv.wait_for(lk, std::chrono::duration<double>(timeoutSec),
                       [&] { return st->done; }))
    MEGBA_CHECK(false,
                "rccl_preflight: timed out after " +
                    std::to_string(timeoutSec) +
                    "s waiting for comm init + allreduce (rank " +
                    std::to_string(rank) + "/" + std::to_string(world) +
                    ") — a peer rank is missing or the rendezvous is wedged");
  MEGBA_CHECK(st->err.empty(), "rccl_preflight: " + st->err);
  return std::chrono::duration<double>(std::chrono::steady_clock::now() - t0)
      .count();
}

std::string rcclUniqueIdString() {
  ncclUniqueId id;
  RCCL_CHECK(ncclGetUniqueId(&id));
  return std::string(reinterpret_cast<const char*>(&id), sizeof(id));
}

void deviceSynchronize() { HIP_CHECK(hipDeviceSynchronize()); }

int hipDeviceCountSafe() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

std::pair<long long, long long> hipMemInfoSafe() {
  size_t freeB = 0, totalB = 0;
  if (hipMemGetInfo(&freeB, &totalB) != hipSuccess) return {0, 0};
  return {(long long)freeB, (long long)totalB};
}

}  // namespace megba
