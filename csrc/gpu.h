// GPU data plane: RCCL over xGMI on a dedicated high-priority HIP stream.
//
// Re-design of the reference's GPUContext/GPUOpContext + NCCL op set
// (horovod/common/ops/gpu_operations.{cc,h} N7, nccl_operations.{cc,h} N8)
// for MI355X: one process per GPU, RCCL as the only backend, torch's HIP
// stream pool for allocator-safe stream interop, pack/unpack through the
// CDNA4 batched-copy kernels (kernels.hip), async completion via a finalizer
// thread.
#pragma once

#include <cstdint>
#include <vector>

#include <ATen/ATen.h>

namespace hvd {
struct GlobalState;
struct Response;
struct TensorTableEntry;

namespace gpu {

// Record an event on torch's current stream for `device` (the producer
// ordering point the comm stream will wait on).  Returns an opaque handle.
uintptr_t RecordReadyEvent(int device);

// Execute a (possibly fused) response on the GPU.  Asynchronous: entries'
// callbacks fire from the finalizer thread once the comm-stream work is done.
void Execute(GlobalState& st, Response& resp,
             std::vector<TensorTableEntry>& entries);

// Drain the finalizer queue (shutdown path).
void WaitAllPending();
void Shutdown();

// Abort every live RCCL communicator (TCP peer loss, async RCCL error, or
// stall shutdown).  Idempotent; unblocks hung collectives so pending
// handles fail with ABORTED (-> HorovodInternalError) instead of hanging.
// Reference: nccl_operations.cc:56-147 commDestroyOrAbort.
void AbortComms(const std::string& why);
// True once AbortComms ran (cleared by Shutdown for elastic re-init).
bool CommsFailed();

// Autotuner hook: adjust the ring-vs-one-shot crossover at runtime
// (applied rank-synchronously from TUNE responses).
void SetOneshotThreshold(int64_t bytes);

// True once a RCCL communicator has been created (used by tests to assert
// the native path ran).
bool RcclUsed();

// Direct Adasum pairwise combine on torch's current stream:
// a[i] = acoef*a[i] + bcoef*b[i] per tensor (kernel unit-test surface; the
// collective path uses the same kernels inside ExecuteAdasum).
void AdasumCombine(std::vector<at::Tensor>& a, std::vector<at::Tensor>& b);

// Fused BatchNorm(+Add)+ReLU (training forward + backward); see
// bn_kernels.hip.  residual may be undefined.
std::vector<at::Tensor> FusedBnReluForward(at::Tensor x, at::Tensor residual,
                                           at::Tensor gamma, at::Tensor beta,
                                           at::Tensor running_mean,
                                           at::Tensor running_var,
                                           double momentum, double eps);
std::vector<at::Tensor> FusedBnReluBackward(at::Tensor x, at::Tensor y,
                                            at::Tensor dy, at::Tensor mean,
                                            at::Tensor invstd, at::Tensor gamma,
                                            bool need_residual_grad);

// Fused SGD step on torch's current stream (one kernel for all buckets;
// empty `momenta` = plain SGD).  fp32 tensors.
void FusedSgdStep(std::vector<at::Tensor>& params,
                  std::vector<at::Tensor>& grads,
                  std::vector<at::Tensor>& momenta, double lr, double momentum,
                  double weight_decay, double dampening, bool nesterov);

// Fused AdamW step (decoupled weight decay, torch.optim.AdamW math);
// step is the 1-based step count for bias correction.  fp32 tensors.
void FusedAdamwStep(std::vector<at::Tensor>& params,
                    std::vector<at::Tensor>& grads,
                    std::vector<at::Tensor>& exp_avgs,
                    std::vector<at::Tensor>& exp_avg_sqs, double lr,
                    double beta1, double beta2, double eps,
                    double weight_decay, int64_t step);

}  // namespace gpu
}  // namespace hvd
