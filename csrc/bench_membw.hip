// bench_membw.hip — standalone pure-read HBM bandwidth probe for the
// search-scan access shape (grid-stride uint4 loads, light VALU test
// per chunk). Establishes the box's read ceiling so k_search can be
// judged against hardware, not guesses.
//
// Build+run (on a GPU box):
//   hipcc --offload-arch=gfx950 -O3 csrc/bench_membw.hip -o /tmp/membw
//   /tmp/membw
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define CHK(x)                                                                \
  do {                                                                        \
    hipError_t e = (x);                                                       \
    if (e != hipSuccess) {                                                    \
      fprintf(stderr, "HIP err %s @%d\n", hipGetErrorString(e), __LINE__);    \
      exit(1);                                                                \
    }                                                                         \
  } while (0)

using u32 = unsigned int;
using u64 = unsigned long long;

__device__ __forceinline__ u32 bem(u32 w, u32 splat) {
  const u32 x = w ^ splat;
  return (x - 0x01010101u) & ~x & 0x80808080u;
}

// P loads in flight per lane, grid-stride; accumulates a test mask so
// nothing is dead-code-eliminated.
template <int P>
__global__ void k_read(const uint4 *__restrict__ src, u64 nchunks,
                       u32 splat, u32 *__restrict__ sink) {
  const u64 stride = (u64)gridDim.x * blockDim.x;
  u64 ci = (u64)blockIdx.x * blockDim.x + threadIdx.x;
  u32 acc = 0;
  for (; ci + (u64)(P - 1) * stride < nchunks; ci += (u64)P * stride) {
    uint4 v[P];
#pragma unroll
    for (int p = 0; p < P; ++p)
      v[p] = src[ci + (u64)p * stride];
#pragma unroll
    for (int p = 0; p < P; ++p)
      acc |= bem(v[p].x, splat) | bem(v[p].y, splat) | bem(v[p].z, splat) |
             bem(v[p].w, splat);
  }
  if (acc == 0xDEADBEEFu) // never true for this data
    atomicAdd(sink, 1u);
}

int main() {
  const size_t bytes = 4ull << 30; // 4 GiB region
  const u64 nchunks = bytes / 16;
  uint4 *d;
  u32 *sink;
  CHK(hipMalloc(&d, bytes));
  CHK(hipMalloc(&sink, 4));
  CHK(hipMemset(d, 0x41, bytes));
  CHK(hipMemset(sink, 0, 4));
  for (int blocks : {2048, 4096, 8192, 16384, 32768}) {
    for (int P : {1, 2, 4, 8}) {
      hipEvent_t a, b;
      CHK(hipEventCreate(&a));
      CHK(hipEventCreate(&b));
      auto launch = [&]() {
        switch (P) {
        case 1: hipLaunchKernelGGL(k_read<1>, dim3(blocks), dim3(256), 0, 0, d, nchunks, 0x4E4E4E4Eu, sink); break;
        case 2: hipLaunchKernelGGL(k_read<2>, dim3(blocks), dim3(256), 0, 0, d, nchunks, 0x4E4E4E4Eu, sink); break;
        case 4: hipLaunchKernelGGL(k_read<4>, dim3(blocks), dim3(256), 0, 0, d, nchunks, 0x4E4E4E4Eu, sink); break;
        case 8: hipLaunchKernelGGL(k_read<8>, dim3(blocks), dim3(256), 0, 0, d, nchunks, 0x4E4E4E4Eu, sink); break;
        }
      };
      launch(); // warm
      CHK(hipDeviceSynchronize());
      CHK(hipEventRecord(a));
      for (int i = 0; i < 5; ++i)
        launch();
      CHK(hipEventRecord(b));
      CHK(hipEventSynchronize(b));
      float ms;
      CHK(hipEventElapsedTime(&ms, a, b));
      printf("blocks=%5d P=%d  %.2f TB/s\n", blocks, P,
             5.0 * bytes / (ms * 1e-3) / 1e12);
      CHK(hipEventDestroy(a));
      CHK(hipEventDestroy(b));
    }
  }
  return 0;
}
