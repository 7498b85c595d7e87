// tools_bwprobe.hip — raw HIP bandwidth calibration for the current box.
// Measures pure-read, pure-write (plain + nontemporal), and copy rates with
// dwordx4 per lane, the same access shape as the gfrs kernels.
//   hipcc --offload-arch=gfx950 -O3 tools_bwprobe.hip -o gpurun_out/bwprobe
#include <hip/hip_runtime.h>

#include <cstdio>

typedef uint32_t u32x4 __attribute__((ext_vector_type(4)));

__global__ void k_read(const uint4 *__restrict__ a, size_t n16, uint4 *sink) {
  uint4 acc{0, 0, 0, 0};
  for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += size_t(gridDim.x) * blockDim.x) {
    const uint4 v = a[i];
    acc.x ^= v.x; acc.y ^= v.y; acc.z ^= v.z; acc.w ^= v.w;
  }
  if (acc.x == 0xDEADBEEF) *sink = acc; // never true; defeats DCE
}

template <bool NT>
__global__ void k_write(uint4 *__restrict__ a, size_t n16) {
  const uint4 v{1, 2, 3, 4};
  for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += size_t(gridDim.x) * blockDim.x) {
    if (NT) {
      u32x4 x = {v.x, v.y, v.z, v.w};
      __builtin_nontemporal_store(x, reinterpret_cast<u32x4 *>(a + i));
    } else {
      a[i] = v;
    }
  }
}

template <bool NT>
__global__ void k_copy(uint4 *__restrict__ d, const uint4 *__restrict__ s,
                       size_t n16) {
  for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += size_t(gridDim.x) * blockDim.x) {
    const uint4 v = s[i];
    if (NT) {
      u32x4 x = {v.x, v.y, v.z, v.w};
      __builtin_nontemporal_store(x, reinterpret_cast<u32x4 *>(d + i));
    } else {
      d[i] = v;
    }
  }
}

/* encode-shaped mix: read 2 streams, write 1 (same ratio as RS(6+3)) */
__global__ void k_mix21(uint4 *__restrict__ d, const uint4 *__restrict__ s0,
                        const uint4 *__restrict__ s1, size_t n16) {
  for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += size_t(gridDim.x) * blockDim.x) {
    const uint4 a = s0[i];
    const uint4 b = s1[i];
    uint4 v{a.x ^ b.x, a.y ^ b.y, a.z ^ b.z, a.w ^ b.w};
    d[i] = v;
  }
}

int main() {
  const size_t bytes = size_t(8) << 30;
  const size_t n16 = bytes / 16;
  uint4 *a, *b;
  hipMalloc(&a, bytes);
  hipMalloc(&b, bytes);
  hipMemset(a, 1, bytes);
  hipMemset(b, 2, bytes);
  dim3 grid(2048), blk(256);
  const int reps = 10;

  // read: (a,b)->(src,sink)
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_read, grid, blk, 0, 0, a, n16, b);
    hipDeviceSynchronize();
    hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_read, grid, blk, 0, 0, a, n16, b);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s\n", "read", double(bytes) / (ms / reps / 1e3) / 1e9);
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_write<false>, grid, blk, 0, 0, a, n16);
    hipDeviceSynchronize(); hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_write<false>, grid, blk, 0, 0, a, n16);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s\n", "write", double(bytes) / (ms / reps / 1e3) / 1e9);
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_write<true>, grid, blk, 0, 0, a, n16);
    hipDeviceSynchronize(); hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_write<true>, grid, blk, 0, 0, a, n16);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s\n", "write_nt", double(bytes) / (ms / reps / 1e3) / 1e9);
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_copy<false>, grid, blk, 0, 0, b, a, n16);
    hipDeviceSynchronize(); hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_copy<false>, grid, blk, 0, 0, b, a, n16);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s (moved)\n", "copy", 2.0 * bytes / (ms / reps / 1e3) / 1e9);
  }
  {
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_copy<true>, grid, blk, 0, 0, b, a, n16);
    hipDeviceSynchronize(); hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_copy<true>, grid, blk, 0, 0, b, a, n16);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s (moved)\n", "copy_nt", 2.0 * bytes / (ms / reps / 1e3) / 1e9);
  }
  {
    uint4 *c;
    hipMalloc(&c, bytes);
    hipMemset(c, 3, bytes);
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipLaunchKernelGGL(k_mix21, grid, blk, 0, 0, c, a, b, n16);
    hipDeviceSynchronize(); hipEventRecord(e0);
    for (int r = 0; r < reps; r++)
      hipLaunchKernelGGL(k_mix21, grid, blk, 0, 0, c, a, b, n16);
    hipEventRecord(e1); hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    printf("%-14s %8.1f GB/s (moved, 2r:1w)\n", "mix21",
           3.0 * bytes / (ms / reps / 1e3) / 1e9);
  }
  return 0;
}
