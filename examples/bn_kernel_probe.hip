// Standalone BN-kernel bandwidth probe (no torch — safe under rocprofv3
// --pmc, which crashes when injected into the full torch stack).
// Times each bn kernel on a given shape and prints achieved GB/s vs the
// bytes each pass must move.  Build:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 -Icsrc \
//         examples/bn_kernel_probe.hip csrc/bn_kernels.hip -o bn_probe
// Run: ./bn_probe [N C H W]   (default 64 64 112 112)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdio>
#include <cstdlib>

#include "kernels.h"

using namespace hvd::gpu;

#define CHECK(c)                                                         \
  do {                                                                   \
    hipError_t e = (c);                                                  \
    if (e != hipSuccess) {                                               \
      std::fprintf(stderr, "FAIL %s: %s\n", #c, hipGetErrorString(e));   \
      std::exit(1);                                                      \
    }                                                                    \
  } while (0)

__global__ void fill_k(__hip_bfloat16* p, long long n) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    p[i] = __float2bfloat16(0.001f * (float)(i % 977));
}

int main(int argc, char** argv) {
  long long N = 64, C = 64, H = 112, W = 112;
  if (argc >= 5) {
    N = atoll(argv[1]);
    C = atoll(argv[2]);
    H = atoll(argv[3]);
    W = atoll(argv[4]);
  }
  const long long total = N * C * H * W;
  const long long count = total / C;
  const long long BANK = (long long)kBnBanks * 2 * C;
  __hip_bfloat16 *x, *y, *dy, *dx;
  float *banks, *params, *banks2;
  CHECK(hipMalloc(&x, total * 2));
  CHECK(hipMalloc(&y, total * 2));
  CHECK(hipMalloc(&dy, total * 2));
  CHECK(hipMalloc(&dx, total * 2));
  CHECK(hipMalloc(&banks, BANK * sizeof(float)));
  CHECK(hipMalloc(&params, 4 * C * sizeof(float)));  // mean,invstd,g,b
  CHECK(hipMalloc(&banks2, BANK * sizeof(float)));
  fill_k<<<1024, 256>>>(x, total);
  fill_k<<<1024, 256>>>(dy, total);
  CHECK(hipMemset(banks, 0, BANK * sizeof(float)));
  CHECK(hipMemset(params, 0, 4 * C * sizeof(float)));
  CHECK(hipDeviceSynchronize());

  hipStream_t s;
  CHECK(hipStreamCreate(&s));
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));

  auto bench = [&](const char* name, double bytes, auto fn) {
    for (int i = 0; i < 3; ++i) fn();
    CHECK(hipStreamSynchronize(s));
    const int iters = 50;
    CHECK(hipEventRecord(e0, s));
    for (int i = 0; i < iters; ++i) fn();
    CHECK(hipEventRecord(e1, s));
    CHECK(hipEventSynchronize(e1));
    float ms = 0;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    double per = ms / iters;
    std::printf("%-14s %8.1f us  %7.0f GB/s (of ~8000)\n", name, per * 1e3,
                bytes / (per * 1e-3) / 1e9);
  };

  const double P = (double)total * 2;  // one bf16 pass over the activation
  bench("stats", P, [&] {
    hipMemsetAsync(banks, 0, BANK * sizeof(float), s);
    BnStatsLaunch(x, total, (int)C, DT_BF16, banks, nullptr, s);
  });
  BnBankReduceLaunch(banks, (int)C, s);
  BnFinalizeLaunch(banks, params, params + C, nullptr, nullptr, count, 0.1f,
                   1e-5f, (int)C, s);
  bench("apply", 2 * P, [&] {
    BnApplyReluLaunch(x, nullptr, y, params, params + C, params + 2 * C,
                      params + 3 * C, total, (int)C, DT_BF16, s);
  });
  bench("bwd_stats", 3 * P, [&] {
    hipMemsetAsync(banks2, 0, BANK * sizeof(float), s);
    BnBwdStatsLaunch(x, y, dy, nullptr, params, params + C, total, (int)C,
                     DT_BF16, banks2, nullptr, s);
    BnBankReduceLaunch(banks2, (int)C, s);
  });
  bench("bwd_apply", 4 * P, [&] {
    BnBwdApplyLaunch(x, y, dy, dx, params, params + C, params + 2 * C,
                     banks2, banks2 + C, total, (int)C, DT_BF16,
                     1.0f / (float)count, s);
  });
  return 0;
}
