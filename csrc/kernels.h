// Host-visible interface to the CDNA4 kernels (kernels.hip).
#pragma once

#include <hip/hip_runtime.h>

namespace hvd {
namespace gpu {

// Matches hvd::DataType codes (common.h) — kept as plain ints so this header
// stays torch-free for the .hip TU.
enum KDType {
  DT_U8 = 0,
  DT_I8 = 1,
  DT_I32 = 2,
  DT_I64 = 3,
  DT_F16 = 4,
  DT_F32 = 5,
  DT_F64 = 6,
  DT_BOOL = 7,
  DT_BF16 = 8,
  DT_U16 = 9,
  DT_I16 = 10,
};

// Descriptor capacity per launch: sized so the by-value kernarg stays well
// under the 4 KB limit (48 * 28 B + 8 ≈ 1.4 KB).
constexpr int kCopyBatchCapacity = 48;

struct CopyBatchArgs {
  const void* src[kCopyBatchCapacity];
  void* dst[kCopyBatchCapacity];
  unsigned long long numel[kCopyBatchCapacity];
  double scale[kCopyBatchCapacity];
  int count = 0;
};

// Batched strided copy with optional per-entry scale and src->dst dtype
// conversion.  grid = count * blocks_per_copy blocks of 256 threads.
hipError_t BatchedCopyLaunch(const CopyBatchArgs& args, int src_dt, int dst_dt,
                             bool with_scale, int blocks_per_copy,
                             hipStream_t stream);

// ---- One-shot xGMI allreduce (oneshot.hip) --------------------------------
// Every rank reads all n peers' staging slots directly over the
// fully-connected xGMI mesh and reduces locally — one step instead of a
// 2(n-1)-hop ring; the win is latency on small/medium buckets.
constexpr int kOneshotMaxRanks = 8;

struct OneshotDeviceArgs {
  // set-local-rank-indexed device pointers, slot-adjusted by the host
  const void* staging[kOneshotMaxRanks];
  void* flags[kOneshotMaxRanks];  // fine-grained uint64 pages
  int n = 0;
  int li = 0;
};

// Spin (1 tiny block) until every rank's consumed counter reaches min_seq —
// my staging slot for the next op is free to overwrite.
hipError_t OneshotWaitConsumedLaunch(const OneshotDeviceArgs& args,
                                     unsigned long long min_seq,
                                     hipStream_t stream);
// Signal ready, wait all ranks, reduce n staging slots into dst (local
// fusion buffer).  bytes must be 16-aligned (fusion padding guarantees it);
// op: 0=sum 1=min 2=max 3=product; dt in {DT_F32,DT_F64,DT_F16,DT_BF16}.
hipError_t OneshotReduceLaunch(const OneshotDeviceArgs& args,
                               unsigned long long seq, int dt,
                               unsigned long long bytes, void* dst, int op,
                               hipStream_t stream);

// ---- Adasum device pipeline (adasum_kernels.hip) ---------------------------
// Stage 1: per-tensor double-precision dot products / squared norms of (a,b)
// pairs; Stage 2: a = acoef*a + bcoef*b with coefficients derived on-device.
struct AdasumBatchArgs {
  void* a[kCopyBatchCapacity];        // dtype = wire dtype
  const void* b[kCopyBatchCapacity];
  unsigned long long numel[kCopyBatchCapacity];
  int count = 0;
};

// dots layout: [count][3] doubles = {dot(a,b), |a|^2, |b|^2}; must be zeroed
// before the dot pass.
hipError_t AdasumDotsLaunch(const AdasumBatchArgs& args, int dt, double* dots,
                            hipStream_t stream);
hipError_t AdasumScaledAddLaunch(const AdasumBatchArgs& args, int dt,
                                 const double* dots, hipStream_t stream);

// ---- Fused SGD (sgd_kernels.hip) ------------------------------------------
struct SgdBatchArgs {
  void* params[kCopyBatchCapacity];
  const void* grads[kCopyBatchCapacity];
  void* momenta[kCopyBatchCapacity];
  unsigned long long numel[kCopyBatchCapacity];
  int count = 0;
};

hipError_t FusedSgdLaunch(const SgdBatchArgs& args, float lr, float momentum,
                          float weight_decay, float dampening, bool nesterov,
                          hipStream_t stream);

// ---- Fused AdamW (adamw_kernels.hip) --------------------------------------
struct AdamwBatchArgs {
  void* params[kCopyBatchCapacity];
  const void* grads[kCopyBatchCapacity];
  void* exp_avg[kCopyBatchCapacity];
  void* exp_avg_sq[kCopyBatchCapacity];
  unsigned long long numel[kCopyBatchCapacity];
  int count = 0;
};

// step is 1-based (bias correction uses beta^step).
hipError_t FusedAdamwLaunch(const AdamwBatchArgs& args, float lr, float beta1,
                            float beta2, float eps, float weight_decay,
                            long long step, hipStream_t stream);

// ---- Fused BatchNorm(+Add)+ReLU (bn_kernels.hip) --------------------------
// NHWC dense activations [total=N*H*W rows, C channels], C % 8 == 0,
// dt in {DT_F32, DT_F16, DT_BF16}; stats/params fp32.
//
// The reduction kernels flush block partials into kBnBanks rows of [2*C]
// floats (zeroed by the caller): one shared accumulator serializes at the
// memory controller under thousands of blocks.
constexpr int kBnBanks = 64;

// sums: banked partial buffer of kBnBanks * 2 * C floats (zeroed).
hipError_t BnStatsLaunch(const void* x, long long total, int C, int dt,
                         float* sums, float* sqs_unused, hipStream_t stream);
// Per-channel epilogue in ONE kernel: bank-reduce + mean/invstd + running
// stats — replaces ~10 tiny ATen launches per BN layer (measured 3.5
// ms/step of pure launch overhead at batch 64).
hipError_t BnFinalizeLaunch(const float* banks, float* mean, float* invstd,
                            float* running_mean, float* running_var,
                            long long count, float momentum, float eps,
                            int C, hipStream_t stream);
// Reduce banks into banks[0..2C) (the bwd tail; fwd folds it into finalize).
hipError_t BnBankReduceLaunch(float* banks, int C, hipStream_t stream);
hipError_t BnApplyReluLaunch(const void* x, const void* res, void* y,
                             const float* mean, const float* invstd,
                             const float* gamma, const float* beta,
                             long long total, int C, int dt,
                             hipStream_t stream);
hipError_t BnBwdStatsLaunch(const void* x, const void* y, const void* dy,
                            void* g_out, const float* mean,
                            const float* invstd, long long total, int C,
                            int dt, float* sum_g, float* sum_gx,
                            hipStream_t stream);
hipError_t BnBwdApplyLaunch(const void* x, const void* y, const void* dy,
                            void* dx, const float* mean, const float* invstd,
                            const float* gamma, const float* sum_g,
                            const float* sum_gx, long long total, int C,
                            int dt, float inv_count, hipStream_t stream);

}  // namespace gpu
}  // namespace hvd
