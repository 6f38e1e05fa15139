// Microbenchmark: per-CU staging rate of global_load_lds vs register
// staging (plain loads + ds_write) in the grouped-GEMM geometry:
// 512 threads / 8 waves, 32 KiB staged per iteration into 128 KiB LDS,
// source L2/L3-resident (16 MiB buffer), barriers per iteration.
// Decides whether the GEMM's ~26 GB/s/CU effective staging is a glds
// engine limit or a schedule artifact.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((address_space(1))) const uint32_t gas_u32;
typedef __attribute__((address_space(3))) uint32_t las_u32;
typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;

// each iteration: 8 glds per wave (8 KiB/wave, 32 KiB/block... scaled up:
// stage 64 KiB per block per iter = 16 glds/wave? keep 8/wave = 64KiB/blk)
__global__ __launch_bounds__(512) void k_glds(const uint32_t* __restrict__ src,
                                              uint32_t* __restrict__ sink,
                                              int iters, int srcDwords) {
  __shared__ __attribute__((aligned(16))) char smem[128 * 1024];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  uint32_t acc = 0;
  // per-wave base offsets; walk the source to stay cache-resident
  int off = ((blockIdx.x * 8 + wave) * 4096 + lane * 4) % (srcDwords - 8192);
  for (int it = 0; it < iters; ++it) {
    const int buf = it & 1;  // 2x 64KB halves
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)(src + off + i * 256),
          (las_u32*)(smem + buf * 65536 + (wave * 8 + i) * 1024), 16, 0, 0);
    }
    off = (off + 2048) % (srcDwords - 8192);
    if (it + 1 < iters) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    // touch one dword so nothing is dead
    acc += *reinterpret_cast<const uint32_t*>(smem + buf * 65536 + tid * 4);
    __builtin_amdgcn_s_barrier();
  }
  if (acc == 0xdeadbeef) sink[tid] = acc;
}

__global__ __launch_bounds__(512) void k_regstage(
    const uint32_t* __restrict__ src, uint32_t* __restrict__ sink, int iters,
    int srcDwords) {
  __shared__ __attribute__((aligned(16))) char smem[128 * 1024];
  const int tid = threadIdx.x;
  uint32_t acc = 0;
  int off = (blockIdx.x * 16384 + tid * 4) % (srcDwords - 65536);
  for (int it = 0; it < iters; ++it) {
    const int buf = it & 1;
    u32x4 v[8];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      v[i] = *reinterpret_cast<const u32x4*>(src + off + i * 2048);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<u32x4*>(smem + buf * 65536 + tid * 16 +
                                i * 8192) = v[i];
    off = (off + 2048) % (srcDwords - 65536);
    __syncthreads();
    acc += *reinterpret_cast<const uint32_t*>(smem + buf * 65536 + tid * 4);
    __syncthreads();
  }
  if (acc == 0xdeadbeef) sink[tid] = acc;
}

int main() {
  const int srcBytes = 16 << 20;
  uint32_t *src, *sink;
  (void)hipMalloc(&src, srcBytes);
  (void)hipMalloc(&sink, 4096);
  (void)hipMemset(src, 1, srcBytes);
  const int iters = 2000, blocks = 256;
  const double bytesPer = (double)blocks * 65536.0 * iters;
  for (int variant = 0; variant < 2; ++variant) {
    for (int rep = 0; rep < 3; ++rep) {
      hipEvent_t e0, e1;
      (void)hipEventCreate(&e0);
      (void)hipEventCreate(&e1);
      (void)hipEventRecord(e0, 0);
      if (variant == 0)
        hipLaunchKernelGGL(k_glds, dim3(blocks), dim3(512), 0, 0, src, sink,
                           iters, srcBytes / 4);
      else
        hipLaunchKernelGGL(k_regstage, dim3(blocks), dim3(512), 0, 0, src,
                           sink, iters, srcBytes / 4);
      (void)hipEventRecord(e1, 0);
      (void)hipEventSynchronize(e1);
      float ms = 0;
      (void)hipEventElapsedTime(&ms, e0, e1);
      printf("%s rep%d: %.2f ms -> %.1f GB/s chip, %.1f GB/s per CU\n",
             variant == 0 ? "glds    " : "regstage", rep, ms,
             bytesPer / (ms * 1e-3) / 1e9,
             bytesPer / (ms * 1e-3) / 1e9 / 256.0);
    }
  }
  return 0;
}
