// Fused BatchNorm(+Add)+ReLU kernels for CDNA4, NHWC (channels_last).
//
// Motivation (measured, profiles/pipeline_study.md): MIOpen's split
// BN-forward / BN-backward plus the separate ReLU / residual-add elementwise
// passes are ~30% of a ResNet-50 bf16 step's GPU time.  Fusing
// normalize+affine+add+relu into one HBM pass per tensor (and the backward
// reductions into one) removes whole read/write passes.
//
// Layout contract: activations are channels_last-dense [count=N*H*W, C]
// with C contiguous and C % 8 == 0 (every ResNet channel width).  Stats
// accumulate in fp32 via per-block LDS bins (C <= 4096 -> <= 32 KB LDS)
// flushed with one global atomicAdd per channel per block (guide G12).
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include <algorithm>

#include "kernels.h"

namespace hvd {
namespace gpu {

namespace {

template <typename T>
__device__ __forceinline__ float to_f(T v) { return (float)v; }
template <>
__device__ __forceinline__ float to_f<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T>
__device__ __forceinline__ T from_f(float v) { return (T)v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// ---- forward stats: per-channel sum and sum-of-squares --------------------
// Fixed-channel decomposition: the host sizes the grid so
// (threads * 8) % C == 0, making every thread revisit the SAME 8 channels
// each grid-stride iteration.  Partials live in registers; one LDS flush
// per thread at block end, one global atomicAdd per (block, channel).
template <typename T>
__global__ __launch_bounds__(256) void bn_stats_k(const T* __restrict__ x,
                                                  long long total, int C,
                                                  float* __restrict__ sums,
                                                  float* __restrict__ sqs) {
  extern __shared__ float lds[];  // [C] sum, [C] sq
  float* lsum = lds;
  float* lsq = lds + C;
  for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) lds[c] = 0.f;
  __syncthreads();
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const long long nvec = total / 8;
  const int c0 = (int)((tid * 8) % C);
  float rsum[8] = {0}, rsq[8] = {0};
  // 2x-unrolled independent loads.  NOTE: measured NO effect on the
  // read-only stream's 1.8 TB/s ceiling (vs 7.5 TB/s for the read+write
  // apply pass with the same loop structure) — the cap is not
  // loads-in-flight; left in place since it costs nothing.  Open question
  // recorded in NOTES.md.
  long long i = tid;
  for (; i + nthreads < nvec; i += 2 * nthreads) {
    union { uint4 u; T e[8]; } v0, v1;
    if (sizeof(T) == 2) {
      v0.u = ((const uint4*)x)[i];
      v1.u = ((const uint4*)x)[i + nthreads];
    } else {
      ((uint4*)&v0)[0] = ((const uint4*)x)[i * 2];
      ((uint4*)&v0)[1] = ((const uint4*)x)[i * 2 + 1];
      ((uint4*)&v1)[0] = ((const uint4*)x)[(i + nthreads) * 2];
      ((uint4*)&v1)[1] = ((const uint4*)x)[(i + nthreads) * 2 + 1];
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f0 = to_f<T>(v0.e[k]), f1 = to_f<T>(v1.e[k]);
      rsum[k] += f0 + f1;
      rsq[k] += f0 * f0 + f1 * f1;
    }
  }
  for (; i < nvec; i += nthreads) {
    union { uint4 u; T e[8]; } v;
    if (sizeof(T) == 2) {
      v.u = ((const uint4*)x)[i];
    } else {
      ((uint4*)&v)[0] = ((const uint4*)x)[i * 2];
      ((uint4*)&v)[1] = ((const uint4*)x)[i * 2 + 1];
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = to_f<T>(v.e[k]);
      rsum[k] += f;
      rsq[k] += f * f;
    }
  }
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    atomicAdd(&lsum[c0 + k], rsum[k]);
    atomicAdd(&lsq[c0 + k], rsq[k]);
  }
  __syncthreads();
  // banked flush: 2048 blocks all hitting the same C addresses serialize at
  // the memory controller (measured 1.4 TB/s effective on the stats pass vs
  // 7.3 TB/s for the atomic-free apply pass); spreading blocks over
  // kBnBanks partial rows cuts per-address contention by the bank count
  float* bsum = sums + (size_t)(blockIdx.x & (kBnBanks - 1)) * 2 * C;
  float* bsq = bsum + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    if (lsum[c] != 0.f || lsq[c] != 0.f) {
      atomicAdd(&bsum[c], lsum[c]);
      atomicAdd(&bsq[c], lsq[c]);
    }
  }
}

// ---- forward apply: y = relu(gamma*xhat + beta [+ residual]) --------------
template <typename T, bool ADD>
__global__ __launch_bounds__(256) void bn_apply_relu_k(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    long long total, int C) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const long long nvec = total / 8;
  // fixed-channel decomposition (see bn_stats_k): per-thread channel params
  // load once into registers
  const int c0 = (int)((tid * 8) % C);
  float rm[8], ri[8], rg[8], rb[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    rm[k] = mean[c0 + k];
    ri[k] = invstd[c0 + k];
    rg[k] = gamma[c0 + k];
    rb[k] = beta[c0 + k];
  }
  for (long long i = tid; i < nvec; i += nthreads) {
    union { uint4 u; T e[8]; } vx, vr, vy;
    if (sizeof(T) == 2) {
      vx.u = ((const uint4*)x)[i];
      if (ADD) vr.u = ((const uint4*)res)[i];
    } else {
      ((uint4*)&vx)[0] = ((const uint4*)x)[i * 2];
      ((uint4*)&vx)[1] = ((const uint4*)x)[i * 2 + 1];
      if (ADD) {
        ((uint4*)&vr)[0] = ((const uint4*)res)[i * 2];
        ((uint4*)&vr)[1] = ((const uint4*)res)[i * 2 + 1];
      }
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = (to_f<T>(vx.e[k]) - rm[k]) * ri[k] * rg[k] + rb[k];
      if (ADD) f += to_f<T>(vr.e[k]);
      vy.e[k] = from_f<T>(f > 0.f ? f : 0.f);
    }
    if (sizeof(T) == 2) {
      ((uint4*)y)[i] = vy.u;
    } else {
      ((uint4*)y)[i * 2] = ((uint4*)&vy)[0];
      ((uint4*)y)[i * 2 + 1] = ((uint4*)&vy)[1];
    }
  }
}

// ---- backward stats: g = dy*(y>0); per-channel sum(g), sum(g*xhat);
//      optionally materialize g (residual gradient for the ADD variant)
template <typename T, bool WRITE_G>
__global__ __launch_bounds__(256) void bn_bwd_stats_k(
    const T* __restrict__ x, const T* __restrict__ y, const T* __restrict__ dy,
    T* __restrict__ g_out, const float* __restrict__ mean,
    const float* __restrict__ invstd, long long total, int C,
    float* __restrict__ sum_g, float* __restrict__ sum_gx) {
  extern __shared__ float lds[];
  float* lg = lds;
  float* lgx = lds + C;
  for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) lds[c] = 0.f;
  __syncthreads();
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const long long nvec = total / 8;
  const int c0 = (int)((tid * 8) % C);
  float rmean[8], rinv[8], racc_g[8] = {0}, racc_gx[8] = {0};
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    rmean[k] = mean[c0 + k];
    rinv[k] = invstd[c0 + k];
  }
  for (long long i = tid; i < nvec; i += nthreads) {
    union { uint4 u; T e[8]; } vx, vy, vd, vg;
    if (sizeof(T) == 2) {
      vx.u = ((const uint4*)x)[i];
      vy.u = ((const uint4*)y)[i];
      vd.u = ((const uint4*)dy)[i];
    } else {
      ((uint4*)&vx)[0] = ((const uint4*)x)[i * 2];
      ((uint4*)&vx)[1] = ((const uint4*)x)[i * 2 + 1];
      ((uint4*)&vy)[0] = ((const uint4*)y)[i * 2];
      ((uint4*)&vy)[1] = ((const uint4*)y)[i * 2 + 1];
      ((uint4*)&vd)[0] = ((const uint4*)dy)[i * 2];
      ((uint4*)&vd)[1] = ((const uint4*)dy)[i * 2 + 1];
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float gv = to_f<T>(vy.e[k]) > 0.f ? to_f<T>(vd.e[k]) : 0.f;
      float xh = (to_f<T>(vx.e[k]) - rmean[k]) * rinv[k];
      racc_g[k] += gv;
      racc_gx[k] += gv * xh;
      if (WRITE_G) vg.e[k] = from_f<T>(gv);
    }
    if (WRITE_G) {
      if (sizeof(T) == 2) {
        ((uint4*)g_out)[i] = vg.u;
      } else {
        ((uint4*)g_out)[i * 2] = ((uint4*)&vg)[0];
        ((uint4*)g_out)[i * 2 + 1] = ((uint4*)&vg)[1];
      }
    }
  }
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    atomicAdd(&lg[c0 + k], racc_g[k]);
    atomicAdd(&lgx[c0 + k], racc_gx[k]);
  }
  __syncthreads();
  float* bg = sum_g + (size_t)(blockIdx.x & (kBnBanks - 1)) * 2 * C;
  float* bgx = bg + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    if (lg[c] != 0.f || lgx[c] != 0.f) {
      atomicAdd(&bg[c], lg[c]);
      atomicAdd(&bgx[c], lgx[c]);
    }
  }
}

// reduce the kBnBanks x [2C] partials into banks[0] (both fwd + bwd tails)
__global__ __launch_bounds__(256) void bn_bank_reduce_k(float* banks,
                                                        int C) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < 2 * C;
       c += gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int b = 0; b < kBnBanks; ++b) acc += banks[(size_t)b * 2 * C + c];
    banks[c] = acc;
  }
}

// ---- backward apply: dx = gamma*invstd*(g - sum_g/n - xhat*sum_gx/n) ------
template <typename T>
__global__ __launch_bounds__(256) void bn_bwd_apply_k(
    const T* __restrict__ x, const T* __restrict__ y, const T* __restrict__ dy,
    T* __restrict__ dx, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ sum_g, const float* __restrict__ sum_gx,
    long long total, int C, float inv_count) {
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const long long nvec = total / 8;
  const int c0 = (int)((tid * 8) % C);
  float rmean[8], rinv[8], rgi[8], rsg[8], rsgx[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    rmean[k] = mean[c0 + k];
    rinv[k] = invstd[c0 + k];
    rgi[k] = gamma[c0 + k] * rinv[k];
    rsg[k] = sum_g[c0 + k] * inv_count;
    rsgx[k] = sum_gx[c0 + k] * inv_count;
  }
  for (long long i = tid; i < nvec; i += nthreads) {
    union { uint4 u; T e[8]; } vx, vy, vd, vo;
    if (sizeof(T) == 2) {
      vx.u = ((const uint4*)x)[i];
      vy.u = ((const uint4*)y)[i];
      vd.u = ((const uint4*)dy)[i];
    } else {
      ((uint4*)&vx)[0] = ((const uint4*)x)[i * 2];
      ((uint4*)&vx)[1] = ((const uint4*)x)[i * 2 + 1];
      ((uint4*)&vy)[0] = ((const uint4*)y)[i * 2];
      ((uint4*)&vy)[1] = ((const uint4*)y)[i * 2 + 1];
      ((uint4*)&vd)[0] = ((const uint4*)dy)[i * 2];
      ((uint4*)&vd)[1] = ((const uint4*)dy)[i * 2 + 1];
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float gv = to_f<T>(vy.e[k]) > 0.f ? to_f<T>(vd.e[k]) : 0.f;
      float xh = (to_f<T>(vx.e[k]) - rmean[k]) * rinv[k];
      float v = rgi[k] * (gv - rsg[k] - xh * rsgx[k]);
      vo.e[k] = from_f<T>(v);
    }
    if (sizeof(T) == 2) {
      ((uint4*)dx)[i] = vo.u;
    } else {
      ((uint4*)dx)[i * 2] = ((uint4*)&vo)[0];
      ((uint4*)dx)[i * 2 + 1] = ((uint4*)&vo)[1];
    }
  }
}

int grid_for(long long total, int C, int vecs_per_thread = 1) {
  // vecs_per_thread > 1 for the REDUCTION kernels: their per-block flush
  // costs 2*C global atomicAdds, so wide-channel shapes with few rows must
  // amortize it over more elements per block (measured: at C=2048, N*H*W
  // = 3136 the old 1-vec grid issued ~1.3 atomics PER ELEMENT and lost
  // ~40% to MIOpen; 8 vecs/thread cuts the flush traffic 8x).  Elementwise
  // kernels keep 1 for maximum bandwidth-filling parallelism.
  long long blocks = (total / 8 + 256LL * vecs_per_thread - 1) /
                     (256LL * vecs_per_thread);
  blocks = std::min<long long>(blocks > 0 ? blocks : 1, 2048);
  // fixed-channel decomposition invariant: (blocks*256*8) % C == 0 so each
  // thread's channel octet is stride-invariant.  vecs_per_row = C/8 <= 512;
  // round blocks up to a multiple of ceil(C/8/256) and ensure divisibility.
  long long vpr = C / 8;  // vectors per row
  long long g = std::__gcd((long long)256, vpr);
  long long mult = vpr / g;  // blocks must be a multiple of this
  blocks = ((blocks + mult - 1) / mult) * mult;
  return (int)blocks;
}

}  // namespace

#define DISPATCH_T(DT, FN)                                        \
  switch (DT) {                                                   \
    case DT_F32: { using scalar_t = float; FN; break; }           \
    case DT_F16: { using scalar_t = __half; FN; break; }          \
    case DT_BF16: { using scalar_t = __hip_bfloat16; FN; break; } \
    default: return hipErrorInvalidValue;                         \
  }

__global__ __launch_bounds__(256) void bn_finalize_k(
    const float* __restrict__ banks, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var, long long count, float momentum,
    float eps, int C) {
  const float inv_n = 1.0f / (float)count;
  const float ub = count > 1 ? (float)count / (float)(count - 1) : 1.0f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    // banks were reduced into row 0 by bn_bank_reduce_k (multi-block; a
    // single-block 64x2C read here cost ~40 us at C=2048)
    float m = banks[c] * inv_n;
    float v = banks[C + c] * inv_n - m * m;
    mean[c] = m;
    invstd[c] = rsqrtf(v + eps);
    if (running_mean) {
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
      running_var[c] =
          (1.f - momentum) * running_var[c] + momentum * v * ub;
    }
  }
}

hipError_t BnFinalizeLaunch(const float* banks, float* mean, float* invstd,
                            float* running_mean, float* running_var,
                            long long count, float momentum, float eps,
                            int C, hipStream_t stream) {
  bn_finalize_k<<<1, 256, 0, stream>>>(banks, mean, invstd, running_mean,
                                       running_var, count, momentum, eps, C);
  return hipGetLastError();
}

hipError_t BnBankReduceLaunch(float* banks, int C, hipStream_t stream) {
  int blocks = (2 * C + 255) / 256;
  bn_bank_reduce_k<<<blocks, 256, 0, stream>>>(banks, C);
  return hipGetLastError();
}

hipError_t BnStatsLaunch(const void* x, long long total, int C, int dt,
                         float* sums, float* sqs, hipStream_t stream) {
  int blocks = grid_for(total, C, 8);
  size_t lds = 2 * (size_t)C * sizeof(float);
  DISPATCH_T(dt, (bn_stats_k<scalar_t><<<blocks, 256, lds, stream>>>(
                     (const scalar_t*)x, total, C, sums, sqs)));
  return hipGetLastError();
}

hipError_t BnApplyReluLaunch(const void* x, const void* res, void* y,
                             const float* mean, const float* invstd,
                             const float* gamma, const float* beta,
                             long long total, int C, int dt,
                             hipStream_t stream) {
  int blocks = grid_for(total, C);
  if (res) {
    DISPATCH_T(dt, (bn_apply_relu_k<scalar_t, true><<<blocks, 256, 0, stream>>>(
                       (const scalar_t*)x, (const scalar_t*)res, (scalar_t*)y,
                       mean, invstd, gamma, beta, total, C)));
  } else {
    DISPATCH_T(dt,
               (bn_apply_relu_k<scalar_t, false><<<blocks, 256, 0, stream>>>(
                   (const scalar_t*)x, nullptr, (scalar_t*)y, mean, invstd,
                   gamma, beta, total, C)));
  }
  return hipGetLastError();
}

hipError_t BnBwdStatsLaunch(const void* x, const void* y, const void* dy,
                            void* g_out, const float* mean,
                            const float* invstd, long long total, int C,
                            int dt, float* sum_g, float* sum_gx,
                            hipStream_t stream) {
  int blocks = grid_for(total, C, 8);
  size_t lds = 2 * (size_t)C * sizeof(float);
  if (g_out) {
    DISPATCH_T(dt,
               (bn_bwd_stats_k<scalar_t, true><<<blocks, 256, lds, stream>>>(
                   (const scalar_t*)x, (const scalar_t*)y, (const scalar_t*)dy,
                   (scalar_t*)g_out, mean, invstd, total, C, sum_g, sum_gx)));
  } else {
    DISPATCH_T(dt,
               (bn_bwd_stats_k<scalar_t, false><<<blocks, 256, lds, stream>>>(
                   (const scalar_t*)x, (const scalar_t*)y, (const scalar_t*)dy,
                   nullptr, mean, invstd, total, C, sum_g, sum_gx)));
  }
  return hipGetLastError();
}

hipError_t BnBwdApplyLaunch(const void* x, const void* y, const void* dy,
                            void* dx, const float* mean, const float* invstd,
                            const float* gamma, const float* sum_g,
                            const float* sum_gx, long long total, int C,
                            int dt, float inv_count, hipStream_t stream) {
  int blocks = grid_for(total, C);
  DISPATCH_T(dt, (bn_bwd_apply_k<scalar_t><<<blocks, 256, 0, stream>>>(
                     (const scalar_t*)x, (const scalar_t*)y,
                     (const scalar_t*)dy, (scalar_t*)dx, mean, invstd, gamma,
                     sum_g, sum_gx, total, C, inv_count)));
  return hipGetLastError();
}

#undef DISPATCH_T

}  // namespace gpu
}  // namespace hvd
