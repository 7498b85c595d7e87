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

/* fused encode+frame shape: 6 read streams, 9 write streams (6 framed
 * copies + 3 parity combos).  MISAL adds the production +4 frame-payload
 * byte offset to every store address; ST: 0 plain, 1 nontemporal, 2 sc1
 * (write-through, drops the line from the XCD L2). */
struct MixPtrs {
  const uint4 *r[6];
  uint8_t *w[9];
};

template <int ST, int MISAL>
__global__ void k_mix69(MixPtrs p, size_t n16) {
  for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += size_t(gridDim.x) * blockDim.x) {
    uint4 v[6];
#pragma unroll
    for (int c = 0; c < 6; c++) v[c] = p.r[c][i];
    uint4 par0{v[0].x ^ v[1].x, v[0].y ^ v[1].y, v[0].z ^ v[1].z,
               v[0].w ^ v[1].w};
    uint4 par1{v[2].x ^ v[3].x, v[2].y ^ v[3].y, v[2].z ^ v[3].z,
               v[2].w ^ v[3].w};
    uint4 par2{v[4].x ^ v[5].x, v[4].y ^ v[5].y, v[4].z ^ v[5].z,
               v[4].w ^ v[5].w};
    const uint4 out[9] = {v[0], v[1], v[2], v[3], v[4],
                          v[5], par0, par1, par2};
#pragma unroll
    for (int c = 0; c < 9; c++) {
      uint8_t *dst = p.w[c] + i * 16 + (MISAL ? 4 : 0);
      if (ST == 1) {
        u32x4 x = {out[c].x, out[c].y, out[c].z, out[c].w};
        __builtin_nontemporal_store(x, reinterpret_cast<u32x4 *>(dst));
      } else if (ST == 2) {
        u32x4 x = {out[c].x, out[c].y, out[c].z, out[c].w};
        asm volatile("global_store_dwordx4 %0, %1, off sc0 sc1"
                     :
                     : "v"(dst), "v"(x)
                     : "memory");
      } else {
        *reinterpret_cast<uint4 *>(dst) = out[c];
      }
    }
  }
}

/* same 6r:9w mix but with PER-BLOCK private stream segments: block b walks
 * its own contiguous L-byte window of each stream (the production shape —
 * every workgroup owns its own stripe's shard/image streams) instead of
 * all blocks sharing one cursor per stream.  The delta vs k_mix69 is the
 * price of many-stream DRAM scatter, which the fused kernel inherits from
 * the workload (independent stripes). */
template <int ST, int RMIS = 0, int WMIS = 0>
__global__ void k_mix69_pb(MixPtrs p, size_t blk_n16) {
  const size_t b0 = size_t(blockIdx.x) * blk_n16;
  for (size_t j = threadIdx.x; j < blk_n16; j += blockDim.x) {
    const size_t i = b0 + j;
    uint4 v[6];
#pragma unroll
    for (int c = 0; c < 6; c++)
      v[c] = *reinterpret_cast<const uint4 *>(
          reinterpret_cast<const uint8_t *>(p.r[c] + i) + (RMIS ? 12 : 0));
    uint4 par0{v[0].x ^ v[1].x, v[0].y ^ v[1].y, v[0].z ^ v[1].z,
               v[0].w ^ v[1].w};
    uint4 par1{v[2].x ^ v[3].x, v[2].y ^ v[3].y, v[2].z ^ v[3].z,
               v[2].w ^ v[3].w};
    uint4 par2{v[4].x ^ v[5].x, v[4].y ^ v[5].y, v[4].z ^ v[5].z,
               v[4].w ^ v[5].w};
    const uint4 out[9] = {v[0], v[1], v[2], v[3], v[4],
                          v[5], par0, par1, par2};
#pragma unroll
    for (int c = 0; c < 9; c++) {
      uint8_t *dst = p.w[c] + i * 16 + (WMIS ? 4 : 0);
      if (ST == 1) {
        u32x4 x = {out[c].x, out[c].y, out[c].z, out[c].w};
        __builtin_nontemporal_store(x, reinterpret_cast<u32x4 *>(dst));
      } else {
        *reinterpret_cast<uint4 *>(dst) = out[c];
      }
    }
  }
}

template <int ST, int RMIS = 0, int WMIS = 0>
static void run_mix69_pb(const char *name, MixPtrs p, size_t n16,
                         size_t bytes) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  dim3 grid(2048), blk(256);
  const size_t blk_n16 = n16 / 2048;
  const int reps = 6;
  hipLaunchKernelGGL((k_mix69_pb<ST, RMIS, WMIS>), grid, blk, 0, 0, p,
                     blk_n16);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int r = 0; r < reps; r++)
    hipLaunchKernelGGL((k_mix69_pb<ST, RMIS, WMIS>), grid, blk, 0, 0, p,
                       blk_n16);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  printf("%-14s %8.1f GB/s (moved, 6r:9w per-block)\n", name,
         15.0 * double(blk_n16) * 2048 * 16 / (ms / reps / 1e3) / 1e9);
}

/* verify-kernel read-pattern probe: one 256-thread workgroup per 64 KiB
 * frame at phase MIS, 4 passes of 4x4096-strided uint4 per lane,
 * grid-strided over frames, two block barriers per frame — the exact
 * access shape of crc32b_verify_reg_k minus the CRC math.  Answers
 * whether that kernel sits at its access pattern's memory ceiling. */
template <int MIS>
__global__ void k_vrfy_read(const uint8_t *__restrict__ a, int64_t total,
                            uint32_t *sink) {
  __shared__ uint32_t red[4];
  uint32_t acc = 0;
  const int lane16 = threadIdx.x * 16;
  for (int64_t fr = blockIdx.x; fr < total; fr += gridDim.x) {
    const uint8_t *pb = a + fr * 65536 + MIS;
    if (threadIdx.x < 4) red[threadIdx.x] = 0;
    __syncthreads();
#pragma unroll
    for (int h = 0; h < 4; h++) {
#pragma unroll
      for (int i = 0; i < 4; i++) {
        const uint4 v = *reinterpret_cast<const uint4 *>(
            pb + h * 16384 + i * 4096 + lane16);
        acc ^= v.x ^ v.y ^ v.z ^ v.w;
      }
    }
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] ^= acc;
    __syncthreads();
  }
  if (acc == 0xDEADBEEFu) *sink = red[0];
}

template <int MIS>
static void run_vrfy_read(const char *name, const uint8_t *a, int64_t total,
                          uint32_t *sink, int grid) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  const int reps = 6;
  hipLaunchKernelGGL((k_vrfy_read<MIS>), dim3(grid), dim3(256), 0, 0, a,
                     total, sink);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int r = 0; r < reps; r++)
    hipLaunchKernelGGL((k_vrfy_read<MIS>), dim3(grid), dim3(256), 0, 0, a,
                       total, sink);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  printf("%-14s %8.1f GB/s (frame-grid read, grid %d)\n", name,
         double(total) * 65536 / (ms / reps / 1e3) / 1e9, grid);
}

template <int ST, int MISAL>
static void run_mix69(const char *name, MixPtrs p, size_t n16, size_t bytes) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  dim3 grid(2048), blk(256);
  const int reps = 6;
  hipLaunchKernelGGL((k_mix69<ST, MISAL>), grid, blk, 0, 0, p, n16);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int r = 0; r < reps; r++)
    hipLaunchKernelGGL((k_mix69<ST, MISAL>), grid, blk, 0, 0, p, n16);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  printf("%-14s %8.1f GB/s (moved, 6r:9w)\n", name,
         15.0 * bytes / (ms / reps / 1e3) / 1e9);
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
  {
    /* 15 streams of 1 GiB: the fused encode+frame mix */
    const size_t sb = size_t(1) << 30;
    const size_t sn16 = sb / 16;
    MixPtrs p{};
    for (int c = 0; c < 6; c++) {
      uint4 *q;
      hipMalloc(&q, sb);
      hipMemset(q, 17 + c, sb);
      p.r[c] = q;
    }
    for (int c = 0; c < 9; c++) {
      uint8_t *q;
      hipMalloc(&q, sb + 16);
      hipMemset(q, 0, sb + 16);
      p.w[c] = q;
    }
    run_mix69<0, 0>("mix69", p, sn16 - 1, sb);
    run_mix69<0, 1>("mix69+4", p, sn16 - 1, sb);
    run_mix69<1, 0>("mix69_nt", p, sn16 - 1, sb);
    run_mix69<1, 1>("mix69_nt+4", p, sn16 - 1, sb);
    run_mix69<2, 0>("mix69_sc1", p, sn16 - 1, sb);
    run_mix69<2, 1>("mix69_sc1+4", p, sn16 - 1, sb);
    run_mix69_pb<0>("mix69_pb", p, sn16, sb);
    run_mix69_pb<1>("mix69_pb_nt", p, sn16, sb);
    /* production misalignment attribution (reads at +12, writes at +4):
     * the streams are 1 GiB + 16 B so the offsets stay in bounds */
    run_mix69_pb<0, 1, 0>("pb_rmis", p, sn16 - 1, sb);
    run_mix69_pb<0, 0, 1>("pb_wmis", p, sn16 - 1, sb);
    run_mix69_pb<0, 1, 1>("pb_rwmis", p, sn16 - 1, sb);
    run_mix69_pb<1, 1, 1>("pb_rwmis_nt", p, sn16 - 1, sb);
    run_mix69_pb<1, 0, 1>("pb_wmis_nt", p, sn16 - 1, sb);
  }
  {
    /* crc32b_verify_reg read-pattern: frames over the first 2 GiB of a
     * (slack for lane 255's +4-phase overhang) */
    const int64_t vt = (int64_t(2) << 30) / 65536;
    uint32_t *sink = reinterpret_cast<uint32_t *>(b);
    const uint8_t *va = reinterpret_cast<const uint8_t *>(a);
    run_vrfy_read<4>("vrfy_rd+4", va, vt, sink, 2048);
    run_vrfy_read<0>("vrfy_rd+0", va, vt, sink, 2048);
    run_vrfy_read<4>("vrfy_rd+4g8", va, vt, sink, 8192);
    run_vrfy_read<4>("vrfy_rd+4gT", va, vt, sink, int(vt));
  }
  return 0;
}
