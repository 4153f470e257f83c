// Probe: cross-process HIP IPC + system-scope atomic flags on MI355X.
// Validates the primitives the one-shot xGMI allreduce (csrc/oneshot.hip)
// depends on: hipIpcGetMemHandle/OpenMemHandle on coarse (hipMalloc) and
// fine-grained (hipExtMallocWithFlags) buffers, plain remote loads after a
// system-scope release flag, and two processes sharing one device (the
// 1-GPU CI box) or two devices (the 8-GPU driver box).
//
// Usage: ipc_probe writer <dir> | ipc_probe reader <dir>
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>
#include <fstream>
#include <thread>
#include <chrono>

#define CHECK(c)                                                          \
  do {                                                                    \
    hipError_t e = (c);                                                   \
    if (e != hipSuccess) {                                                \
      std::fprintf(stderr, "FAIL %s: %s\n", #c, hipGetErrorString(e));    \
      return 1;                                                           \
    }                                                                     \
  } while (0)

__global__ void fill_and_signal(float* data, unsigned long long* flag,
                                int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) data[i] = 2.0f * i + 1.0f;
  __threadfence_system();
  if (i == 0)
    __hip_atomic_store(flag, 42ull, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_SYSTEM);
}

__global__ void wait_and_check(const float* peer_data,
                               unsigned long long* peer_flag, int n,
                               int* errors) {
  if (threadIdx.x == 0) {  // every block waits; barrier below releases it
    while (__hip_atomic_load(peer_flag, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_SYSTEM) != 42ull) {
      __builtin_amdgcn_s_sleep(8);
    }
  }
  __syncthreads();
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n && peer_data[i] != 2.0f * i + 1.0f) atomicAdd(errors, 1);
}

int main(int argc, char** argv) {
  if (argc < 3) return 2;
  const bool writer = std::strcmp(argv[1], "writer") == 0;
  std::string dir = argv[2];
  int ndev = 0;
  CHECK(hipGetDeviceCount(&ndev));
  CHECK(hipSetDevice(writer ? 0 : (ndev > 1 ? 1 : 0)));
  const int N = 1 << 18;

  if (writer) {
    float* data = nullptr;
    unsigned long long* flag = nullptr;
    CHECK(hipMalloc(&data, N * sizeof(float)));  // coarse staging
    CHECK(hipExtMallocWithFlags((void**)&flag, 4096,
                                hipDeviceMallocFinegrained));
    CHECK(hipMemset((void*)flag, 0, 4096));
    hipIpcMemHandle_t hd, hf;
    CHECK(hipIpcGetMemHandle(&hd, data));
    CHECK(hipIpcGetMemHandle(&hf, (void*)flag));
    {
      std::ofstream f(dir + "/handles.bin", std::ios::binary);
      f.write((char*)&hd, sizeof(hd));
      f.write((char*)&hf, sizeof(hf));
    }
    std::ofstream(dir + "/ready").put('1');
    // give the reader a head start so the spin-wait is actually exercised
    std::this_thread::sleep_for(std::chrono::milliseconds(300));
    hipLaunchKernelGGL(fill_and_signal, dim3((N + 255) / 256), dim3(256), 0,
                       0, data, flag, N);
    CHECK(hipDeviceSynchronize());
    // wait for reader verdict
    for (int i = 0; i < 200; ++i) {
      std::ifstream v(dir + "/verdict");
      if (v.good()) {
        std::string s;
        v >> s;
        std::printf("WRITER_SEES_VERDICT %s\n", s.c_str());
        return s == "OK" ? 0 : 1;
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    std::printf("WRITER_TIMEOUT\n");
    return 1;
  }

  // reader
  for (int i = 0; i < 200; ++i) {
    std::ifstream r(dir + "/ready");
    if (r.good()) break;
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
  }
  hipIpcMemHandle_t hd, hf;
  {
    std::ifstream f(dir + "/handles.bin", std::ios::binary);
    if (!f.good()) {
      std::printf("READER_NO_HANDLES\n");
      return 1;
    }
    f.read((char*)&hd, sizeof(hd));
    f.read((char*)&hf, sizeof(hf));
  }
  void *data = nullptr, *flag = nullptr;
  CHECK(hipIpcOpenMemHandle(&data, hd, hipIpcMemLazyEnablePeerAccess));
  CHECK(hipIpcOpenMemHandle(&flag, hf, hipIpcMemLazyEnablePeerAccess));
  int* errors = nullptr;
  CHECK(hipMalloc(&errors, sizeof(int)));
  CHECK(hipMemset(errors, 0, sizeof(int)));
  hipLaunchKernelGGL(wait_and_check, dim3((N + 255) / 256), dim3(256), 0, 0,
                     (const float*)data, (unsigned long long*)flag, N,
                     errors);
  CHECK(hipDeviceSynchronize());
  int herr = -1;
  CHECK(hipMemcpy(&herr, errors, sizeof(int), hipMemcpyDeviceToHost));
  std::printf("READER_ERRORS %d\n", herr);
  std::ofstream(dir + "/verdict") << (herr == 0 ? "OK" : "BAD");
  return herr == 0 ? 0 : 1;
}
