// Microbenchmark: what bounds the grouped-GEMM inner loop on gfx950?
// Three kernels in the EXACT real-kernel geometry (512 threads, 8 waves,
// wave = 2M x 4N, MI=8 A-fragments, NF=4 B-fragments, BK=64, LDS-resident
// operands, barrier per "tile" of 2 s-steps):
//   1. mfma_max:   no LDS reads at all - pure v_mfma_f32_16x16x32_bf16
//                  issue rate at 2 waves/SIMD (the pipe ceiling).
//   2. lds_plain:  12 ds_read_b128 + 32 MFMAs per s-step, plain HIP with
//                  the round-robin + sched_barrier schedule of the real
//                  kernel (compiler-inserted lgkmcnt waits).
//   3. lds_asm:    same reads/MFMAs but ds_read_b128 issued from inline
//                  asm with HAND-COUNTED s_waitcnt lgkmcnt(N) between
//                  fragment groups (the compiler never sees the loads, so
//                  it cannot over-wait).
// If (1) ~== (2), the MFMA pipe itself is the wall and no schedule can
// help. If (3) >> (2), the compiler's waits are the wall and the real
// kernel should move to asm reads. Decides round-2's first work item.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define MFMA(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

__global__ __launch_bounds__(512) void k_mfma_max(const bf16* __restrict__ seed,
                                                  float* __restrict__ sink,
                                                  int iters) {
  const int tid = threadIdx.x;
  bf16x8 af[8], bf[4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    af[i] = *reinterpret_cast<const bf16x8*>(seed + (tid & 63) * 8 + i * 512);
#pragma unroll
  for (int i = 0; i < 4; ++i)
    bf[i] = *reinterpret_cast<const bf16x8*>(seed + 4096 + (tid & 63) * 8 + i * 512);
  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = MFMA(af[mi], bf[ni], acc[mi][ni]);
    }
    __builtin_amdgcn_s_barrier();
  }
  float v = 0;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) v += acc[i][j][0] + acc[i][j][3];
  if (v == 1234.5678f) sink[tid] = v;
}

// shared inner-loop scaffolding for the two LDS variants: A image
// [256][64] bf16 (32 KB) + B image [256][64] bf16 (32 KB), source-chunk
// swizzle exactly as the real kernel (read addr chunk ^= row&7).
template <bool ASM>
__global__ __launch_bounds__(512) void k_lds_mfma(const bf16* __restrict__ seed,
                                                  float* __restrict__ sink,
                                                  int iters) {
  constexpr int BK = 64;
  __shared__ __attribute__((aligned(16))) char smem[2 * 256 * BK * 2];
  bf16* Al = reinterpret_cast<bf16*>(smem);
  bf16* Bl = Al + 256 * BK;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // fill LDS once
  for (int i = tid; i < 2 * 256 * BK / 8; i += 512)
    reinterpret_cast<bf16x8*>(smem)[i] =
        *reinterpret_cast<const bf16x8*>(seed + (i % 4096) * 8);
  __syncthreads();

  const int wr = wave >> 2, wc = wave & 3;
  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int rl = lane & 15;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int cbase = 4 * s + (lane >> 4);
      bf16x8 af[8], bfr[4];
      if constexpr (!ASM) {
        // plain HIP: the real kernel's round-robin schedule
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int R = wc * 64 + ni * 16 + rl;
          bfr[ni] = *reinterpret_cast<const bf16x8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        auto aread = [&](int mi) {
          const int R = wr * 128 + mi * 16 + rl;
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        };
        aread(0); aread(1);
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g2 = 0; g2 < 4; ++g2) {
          if (g2 < 3) { aread(2 * g2 + 2); aread(2 * g2 + 3); }
#pragma unroll
          for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
              acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      } else {
        // asm reads + hand-counted waits. Issue order: b0..b3, a0..a7
        // (12 reads in flight), then per-A-fragment: wait for it and
        // run its 4 MFMAs.
        uint32_t ba[4], aa[8];
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int R = wc * 64 + ni * 16 + rl;
          ba[ni] = (uint32_t)(size_t)(&Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
#pragma unroll
        for (int mi = 0; mi < 8; ++mi) {
          const int R = wr * 128 + mi * 16 + rl;
          aa[mi] = (uint32_t)(size_t)(&Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        __builtin_amdgcn_sched_barrier(0);
        asm volatile(
            "ds_read_b128 %0, %12\n\t"
            "ds_read_b128 %1, %13\n\t"
            "ds_read_b128 %2, %14\n\t"
            "ds_read_b128 %3, %15\n\t"
            "ds_read_b128 %4, %16\n\t"
            "ds_read_b128 %5, %17\n\t"
            "ds_read_b128 %6, %18\n\t"
            "ds_read_b128 %7, %19\n\t"
            "ds_read_b128 %8, %20\n\t"
            "ds_read_b128 %9, %21\n\t"
            "ds_read_b128 %10, %22\n\t"
            "ds_read_b128 %11, %23"
            : "=v"(bfr[0]), "=v"(bfr[1]), "=v"(bfr[2]), "=v"(bfr[3]),
              "=v"(af[0]), "=v"(af[1]), "=v"(af[2]), "=v"(af[3]),
              "=v"(af[4]), "=v"(af[5]), "=v"(af[6]), "=v"(af[7])
            : "v"(ba[0]), "v"(ba[1]), "v"(ba[2]), "v"(ba[3]),
              "v"(aa[0]), "v"(aa[1]), "v"(aa[2]), "v"(aa[3]),
              "v"(aa[4]), "v"(aa[5]), "v"(aa[6]), "v"(aa[7]));
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int mi = 0; mi < 8; ++mi) {
          // b0-b3 and a0..a_mi ready when lgkmcnt <= 7 - mi
          switch (mi) {
            case 0: asm volatile("s_waitcnt lgkmcnt(7)" ::: "memory"); break;
            case 1: asm volatile("s_waitcnt lgkmcnt(6)" ::: "memory"); break;
            case 2: asm volatile("s_waitcnt lgkmcnt(5)" ::: "memory"); break;
            case 3: asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory"); break;
            case 4: asm volatile("s_waitcnt lgkmcnt(3)" ::: "memory"); break;
            case 5: asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory"); break;
            case 6: asm volatile("s_waitcnt lgkmcnt(1)" ::: "memory"); break;
            case 7: asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory"); break;
          }
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      }
    }
    __builtin_amdgcn_s_barrier();
  }
  float v = 0;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) v += acc[i][j][0] + acc[i][j][3];
  if (v == 1234.5678f) sink[tid] = v;
}

// Variant 4: the REAL kernel's staged loop - 256x256 tile, glds
// double/triple-buffered staging from a large (L2+HBM) source, the
// one-barrier stage-first schedule, counted or full vmcnt waits.
// Identical MFMA/read stream to k_lds_mfma. Measures what the staging
// interaction costs vs the LDS-resident 84% ceiling.
typedef __attribute__((address_space(1))) const uint32_t gas_u32p;
typedef __attribute__((address_space(3))) uint32_t las_u32p;

template <int STAGES, bool COUNTED, int BN = 256, bool SCATTER = false,
          int BIGB = 0, bool EPI = false>  // BIGB 2 = rotate panels per iter (cold B)
__global__ __launch_bounds__(512) void k_stage_gemm(const bf16* __restrict__ A,
                                                    const bf16* __restrict__ B,
                                                    float* __restrict__ sink,
                                                    int nK, int iters) {
  constexpr int BM = 256, BK = 64;
  constexpr int NF = BN / 64;
  constexpr int GPW_A = 2, GPW_B = BN / 128, GPT = GPW_A + GPW_B;
  __shared__ __attribute__((aligned(16))) char smem[
      STAGES * (BM + BN) * BK * 2];
  bf16* Abase = reinterpret_cast<bf16*>(smem);
  bf16* Bbase = Abase + STAGES * BM * BK;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int K = nK * BK;
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  const bf16* aSrc[GPW_A];
  const bf16* bSrc[GPW_B];
  int bRow[GPW_B];
#pragma unroll
  for (int i = 0; i < GPW_A; ++i) {
    const int row = (wave * GPW_A + i) * 8 + grow8;
    // SCATTER mimics the token gather of the up GEMM: rows land at
    // pseudo-random positions in a 4096-row x matrix, so each k-tile
    // stages 128B per row at 2 KB stride instead of dense panels
    const size_t arow = SCATTER
        ? (size_t)(((blockIdx.x * BM + row) * 997u) & 4095u)
        : (size_t)blockIdx.x * BM + row;
    aSrc[i] = A + arow * K + schunk * 8;
  }
#pragma unroll
  for (int i = 0; i < GPW_B; ++i) {
    const int row = (wave * GPW_B + i) * 8 + grow8;
    // BIGB mimics the real weight working set: each group of 4 blocks
    // shares one 256-row panel of a 32 MB weight array streamed from HBM
    const size_t panel = BIGB ? (size_t)((blockIdx.x / 4) % 64)
                              : (size_t)(blockIdx.x % 8);
    bSrc[i] = B + (panel * BN + row) * K + schunk * 8;
    bRow[i] = row;
  }
  auto stage = [&](int kt, int buf) {
#pragma unroll
    for (int i = 0; i < GPW_A; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32p*)(aSrc[i] + kt),
          (las_u32p*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 512), 16, 0, 0);
#pragma unroll
    for (int i = 0; i < GPW_B; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32p*)(bSrc[i] + kt),
          (las_u32p*)(Bbase + buf * BN * BK + (wave * GPW_B + i) * 512), 16, 0, 0);
  };
  const int wr = wave >> 2, wc = wave & 3;
  f32x4 acc[8][NF];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
  const int rl = lane & 15;
  for (int it = 0; it < iters; ++it) {
    if constexpr (BIGB == 2) {
      // cold-B: each iteration reads a different panel set, so B always
      // misses L2 (mimics per-forward cold weight streaming)
#pragma unroll
      for (int i = 0; i < GPW_B; ++i) {
        const size_t panel = ((size_t)(blockIdx.x / 4) + (size_t)it * 7) % 64;
        bSrc[i] = B + (panel * BN + bRow[i]) * K + schunk * 8;
      }
    }
    stage(0, 0);
    if constexpr (STAGES == 3) {
      if (nK > 1) stage(BK, 1);
      if constexpr (GPT == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    for (int t = 0; t < nK; ++t) {
      const int pre = t + STAGES - 1;
      const bool stageNow = pre < nK;
      const bool late = (STAGES == 2) && wave >= 4;
      if (stageNow && !late) stage(pre * BK, pre % STAGES);
      const bf16* Al = Abase + (t % STAGES) * BM * BK;
      const bf16* Bl = Bbase + (t % STAGES) * BN * BK;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        if (ss == 1 && late && stageNow) {
          __builtin_amdgcn_s_setprio(0);
          stage(pre * BK, pre % STAGES);
          __builtin_amdgcn_s_setprio(1);
        }
        bf16x8 af[8], bfr[NF];
        const int cbase = 4 * ss + (lane >> 4);
        auto aread = [&](int mi) {
          const int R = wr * 128 + mi * 16 + rl;
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        };
#pragma unroll
        for (int ni = 0; ni < NF; ++ni) {
          const int R = wc * (BN / 4) + ni * 16 + rl;
          bfr[ni] = *reinterpret_cast<const bf16x8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        aread(0); aread(1);
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g2 = 0; g2 < 4; ++g2) {
          if (g2 < 3) { aread(2 * g2 + 2); aread(2 * g2 + 3); }
#pragma unroll
          for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < NF; ++ni)
              acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      if constexpr (STAGES == 3 && COUNTED) {
        if (stageNow) {
          if constexpr (GPT == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
          else asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
        } else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }
  if constexpr (EPI) {
    // the real PHASE-0 epilogue shape: activation + bf16 store of the
    // 256xBN block tile to a global intermediate, bounds-checked
    const int cl = lane & 15;
    const int r0 = (lane >> 4) * 4;
    bf16* out = reinterpret_cast<bf16*>(sink);
    const int routed = 256 * 200;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = wr * 128 + mi * 16 + r0 + r;
        const int m = blockIdx.x * BM + row;
        if (m >= routed) continue;
#pragma unroll
        for (int ni = 0; ni < NF; ++ni) {
          const int col = wc * (BN / 4) + ni * 16 + cl;
          if (col >= BN) continue;
          float vv = acc[mi][ni][r & 3];
          vv = fmaxf(vv, 0.0f);
          out[((size_t)m % 2048) * BN + col] = (bf16)vv;  // sink is 1 MB
        }
      }
    }
    return;
  }
  float v = 0;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) v += acc[i][j][0] + acc[i][j][3];
  if (v == 1234.5678f) sink[tid] = v;
}

// Variant 5: the FULL real up-kernel structure on synthetic data - 3D
// grid (mT,nT,E), XCD remap, sTps token gather from LDS, routed guard,
// per-expert B, strided xM epilogue with activation. iters repeats the
// K loop + epilogue. If this matches the real kernel's ~950 TF, the gap
// vs variant 4 (1689 TF) lives in code we can bisect; if it stays fast,
// the gap is host-side context.
struct MiniTPS { uint32_t tokenIdx; float probSum; };

__global__ __launch_bounds__(512, 2) void k_stage_real(
    const bf16* __restrict__ x, const bf16* __restrict__ W,
    bf16* __restrict__ xM, const MiniTPS* __restrict__ tokenIds,
    const uint32_t* __restrict__ eC, int K, int N, int pEC, int iters) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int GPW_A = 2, GPW_B = 2;
  __shared__ __attribute__((aligned(16))) char smem[
      2 * BM * BK * 2 + 2 * BN * BK * 2 + BM * 8 + 16];
  bf16* Abase = reinterpret_cast<bf16*>(smem);
  bf16* Bbase = Abase + 2 * BM * BK;
  MiniTPS* sTps = reinterpret_cast<MiniTPS*>(Bbase + 2 * BN * BK);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);

  const int mT = gridDim.x, nT = gridDim.y;
  const int nBlocks = mT * nT * gridDim.z;
  const int lin = blockIdx.x + mT * (blockIdx.y + nT * blockIdx.z);
  const int qx = nBlocks / 8, rx = nBlocks % 8;
  const int xcd = lin % 8, pos = lin / 8;
  const int swz =
      (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
  const int e = swz / (mT * nT);
  const int rem = swz % (mT * nT);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int m0 = (rem % mT) * BM;
  const int n0 = (rem / mT) * BN;

  const MiniTPS* tpsE = tokenIds + (size_t)e * pEC;
  if (tid == 0) *sRouted = eC[e];
  __syncthreads();
  const uint32_t routed = *sRouted;
  if ((uint32_t)m0 >= routed) return;
  if (tid < BM) {
    MiniTPS t{0u, 1.0f};
    if ((uint32_t)(m0 + tid) < routed) t = tpsE[m0 + tid];
    sTps[tid] = t;
  }
  __syncthreads();

  const bf16* __restrict__ Bg = W + (size_t)e * 2 * (size_t)N * K;
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  const bf16* aSrc[GPW_A];
  const bf16* bSrc[GPW_B];
#pragma unroll
  for (int i = 0; i < GPW_A; ++i) {
    const int row = (wave * GPW_A + i) * 8 + grow8;
    const size_t arow = (size_t)(sTps[row].tokenIdx & 0x0FFFFFFF);
    aSrc[i] = x + arow * K + schunk * 8;
  }
#pragma unroll
  for (int i = 0; i < GPW_B; ++i) {
    const int row = (wave * GPW_B + i) * 8 + grow8;
    bSrc[i] = Bg + (size_t)min(n0 + row, N - 1) * K + schunk * 8;
  }
  auto stage = [&](int kt, int buf) {
#pragma unroll
    for (int i = 0; i < GPW_A; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32p*)(aSrc[i] + kt),
          (las_u32p*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 512), 16, 0, 0);
#pragma unroll
    for (int i = 0; i < GPW_B; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32p*)(bSrc[i] + kt),
          (las_u32p*)(Bbase + buf * BN * BK + (wave * GPW_B + i) * 512), 16, 0, 0);
  };
  const int wr = wave >> 2, wc = wave & 3;
  f32x4 acc[8][4];
  const int rl = lane & 15;
  const int nK = K / BK;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    for (int t = 0; t < nK; ++t) {
      const bool stageNow = t + 1 < nK;
      const bool late = wave >= 4;
      if (stageNow && !late) stage((t + 1) * BK, (t + 1) & 1);
      const bf16* Al = Abase + (t & 1) * BM * BK;
      const bf16* Bl = Bbase + (t & 1) * BN * BK;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        if (ss == 1 && late && stageNow) {
          __builtin_amdgcn_s_setprio(0);
          stage((t + 1) * BK, (t + 1) & 1);
          __builtin_amdgcn_s_setprio(1);
        }
        bf16x8 af[8], bfr[4];
        const int cbase = 4 * ss + (lane >> 4);
        auto aread = [&](int mi) {
          const int R = wr * 128 + mi * 16 + rl;
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        };
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int R = wc * 64 + ni * 16 + rl;
          bfr[ni] = *reinterpret_cast<const bf16x8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        aread(0); aread(1);
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g2 = 0; g2 < 4; ++g2) {
          if (g2 < 3) { aread(2 * g2 + 2); aread(2 * g2 + 3); }
#pragma unroll
          for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
              acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    // real PHASE-0 epilogue: relu + strided xM store
    const int cl = lane & 15;
    const int r0 = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = wr * 128 + mi * 16 + r0 + r;
        const int m = m0 + row;
        if ((uint32_t)m >= routed) continue;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int col = n0 + wc * 64 + ni * 16 + cl;
          if (col >= N) continue;
          float v = acc[mi][ni][r];
          v = fmaxf(v, 0.0f);
          xM[(size_t)e * pEC * N + (size_t)m * N + col] = (bf16)v;
        }
      }
    }
  }
}

// Variant 6: the real-structure GEMM driven by a TASK QUEUE instead of
// the blockIdx job decomposition - each block claims (e, m0, n0) tile
// descriptors with a device-scope atomic and dereferences them. Compares
// against k_stage_real (same tiles, same order) to price per-tile task
// dispatch for the round-2 fused kernel.
struct TileTask { uint32_t e, m0, n0, pad; };

__global__ __launch_bounds__(512, 2) void k_stage_queued(
    const bf16* __restrict__ x, const bf16* __restrict__ W,
    bf16* __restrict__ xM, const MiniTPS* __restrict__ tokenIds,
    const uint32_t* __restrict__ eC, const TileTask* __restrict__ tasks,
    unsigned long long* claim, int nTasksTotal, int nTaskList, int K,
    int N, int pEC) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int GPW_A = 2, GPW_B = 2;
  __shared__ __attribute__((aligned(16))) char smem[
      2 * BM * BK * 2 + 2 * BN * BK * 2 + BM * 8 + 64];
  bf16* Abase = reinterpret_cast<bf16*>(smem);
  bf16* Bbase = Abase + 2 * BM * BK;
  MiniTPS* sTps = reinterpret_cast<MiniTPS*>(Bbase + 2 * BN * BK);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);
  uint32_t* sTask = sRouted + 1;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rl = lane & 15;
  const int wr = wave >> 2, wc = wave & 3;
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  const int nK = K / BK;
  while (true) {
    __syncthreads();
    if (tid == 0)
      *sTask = (uint32_t)atomicAdd(claim, 1ull);
    __syncthreads();
    const uint32_t tix = *sTask;
    if (tix >= (uint32_t)nTasksTotal) return;
    const TileTask tk = tasks[tix % (uint32_t)nTaskList];
    const int e = tk.e, m0 = tk.m0, n0 = tk.n0;
    const MiniTPS* tpsE = tokenIds + (size_t)e * pEC;
    if (tid == 0) *sRouted = eC[e];
    __syncthreads();
    const uint32_t routed = *sRouted;
    if ((uint32_t)m0 >= routed) continue;
    if (tid < BM) {
      MiniTPS t{0u, 1.0f};
      if ((uint32_t)(m0 + tid) < routed) t = tpsE[m0 + tid];
      sTps[tid] = t;
    }
    __syncthreads();
    const bf16* __restrict__ Bg = W + (size_t)e * 2 * (size_t)N * K;
    const bf16* aSrc[GPW_A];
    const bf16* bSrc[GPW_B];
#pragma unroll
    for (int i = 0; i < GPW_A; ++i) {
      const int row = (wave * GPW_A + i) * 8 + grow8;
      const size_t arow = (size_t)(sTps[row].tokenIdx & 0x0FFFFFFF);
      aSrc[i] = x + arow * K + schunk * 8;
    }
#pragma unroll
    for (int i = 0; i < GPW_B; ++i) {
      const int row = (wave * GPW_B + i) * 8 + grow8;
      bSrc[i] = Bg + (size_t)min(n0 + row, N - 1) * K + schunk * 8;
    }
    auto stage = [&](int kt, int buf) {
#pragma unroll
      for (int i = 0; i < GPW_A; ++i)
        __builtin_amdgcn_global_load_lds(
            (gas_u32p*)(aSrc[i] + kt),
            (las_u32p*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 512), 16, 0, 0);
#pragma unroll
      for (int i = 0; i < GPW_B; ++i)
        __builtin_amdgcn_global_load_lds(
            (gas_u32p*)(bSrc[i] + kt),
            (las_u32p*)(Bbase + buf * BN * BK + (wave * GPW_B + i) * 512), 16, 0, 0);
    };
    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    for (int t = 0; t < nK; ++t) {
      const bool stageNow = t + 1 < nK;
      const bool late = wave >= 4;
      if (stageNow && !late) stage((t + 1) * BK, (t + 1) & 1);
      const bf16* Al = Abase + (t & 1) * BM * BK;
      const bf16* Bl = Bbase + (t & 1) * BN * BK;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        if (ss == 1 && late && stageNow) {
          __builtin_amdgcn_s_setprio(0);
          stage((t + 1) * BK, (t + 1) & 1);
          __builtin_amdgcn_s_setprio(1);
        }
        bf16x8 af[8], bfr[4];
        const int cbase = 4 * ss + (lane >> 4);
        auto aread = [&](int mi) {
          const int R = wr * 128 + mi * 16 + rl;
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        };
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int R = wc * 64 + ni * 16 + rl;
          bfr[ni] = *reinterpret_cast<const bf16x8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        aread(0); aread(1);
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g2 = 0; g2 < 4; ++g2) {
          if (g2 < 3) { aread(2 * g2 + 2); aread(2 * g2 + 3); }
#pragma unroll
          for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
              acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    const int cl = lane & 15;
    const int r0 = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = wr * 128 + mi * 16 + r0 + r;
        const int m = m0 + row;
        if ((uint32_t)m >= routed) continue;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int col = n0 + wc * 64 + ni * 16 + cl;
          if (col >= N) continue;
          float v = acc[mi][ni][r];
          v = fmaxf(v, 0.0f);
          xM[(size_t)e * pEC * N + (size_t)m * N + col] = (bf16)v;
        }
      }
    }
  }
}

// Variant 7: dynamic producer->consumer pipeline. 32 producer blocks
// simulate route tiles (a fixed busy-delay, then emit their share of the
// 512 tile tasks with release-tagged seq); 480 consumer blocks claim and
// compute tiles as they appear. Measures whether the producer phase
// hides behind GEMM fill (the fused kernel's core overlap claim) and
// whether acquire-spin consumers cost anything once tasks flow.
__global__ __launch_bounds__(512, 2) void k_stage_pipe(
    const bf16* __restrict__ x, const bf16* __restrict__ W,
    bf16* __restrict__ xM, const MiniTPS* __restrict__ tokenIds,
    const uint32_t* __restrict__ eC, const TileTask* __restrict__ taskSrc,
    TileTask* ring, uint32_t* seq, unsigned long long* claim,
    int nProducers, int J, int delayLoops, int K, int N, int pEC) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int GPW_A = 2, GPW_B = 2;
  __shared__ __attribute__((aligned(16))) char smem[
      2 * BM * BK * 2 + 2 * BN * BK * 2 + BM * 8 + 64];
  bf16* Abase = reinterpret_cast<bf16*>(smem);
  bf16* Bbase = Abase + 2 * BM * BK;
  MiniTPS* sTps = reinterpret_cast<MiniTPS*>(Bbase + 2 * BN * BK);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);
  uint32_t* sTask = sRouted + 1;
  const int tid = threadIdx.x;
  if ((int)blockIdx.x < nProducers) {
    // producer: busy-wait (simulated route tile), then emit my tasks
    if (tid != 0) return;
    for (volatile int i = 0; i < delayLoops; ++i) { }
    const int per = J / nProducers;
    for (int i = 0; i < per; ++i) {
      const int t = blockIdx.x * per + i;
      ring[t] = taskSrc[t];
      __hip_atomic_store(seq + t, 1u, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
    return;
  }
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rl = lane & 15;
  const int wr = wave >> 2, wc = wave & 3;
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  const int nK = K / BK;
  while (true) {
    __syncthreads();
    if (tid == 0) {
      const uint32_t tix = (uint32_t)atomicAdd(claim, 1ull);
      if (tix < (uint32_t)J) {
        uint32_t spins = 0;
        while (__hip_atomic_load(seq + tix, __ATOMIC_ACQUIRE,
                                 __HIP_MEMORY_SCOPE_AGENT) == 0u) {
          if (++spins > (1u << 26)) break;
          __builtin_amdgcn_s_sleep(1);
        }
      }
      *sTask = tix;
    }
    __syncthreads();
    const uint32_t tix = *sTask;
    if (tix >= (uint32_t)J) return;
    const TileTask tk = ring[tix];
    const int e = tk.e, m0 = tk.m0, n0 = tk.n0;
    const MiniTPS* tpsE = tokenIds + (size_t)e * pEC;
    if (tid == 0) *sRouted = eC[e];
    __syncthreads();
    const uint32_t routed = *sRouted;
    if ((uint32_t)m0 >= routed) continue;
    if (tid < BM) {
      MiniTPS t{0u, 1.0f};
      if ((uint32_t)(m0 + tid) < routed) t = tpsE[m0 + tid];
      sTps[tid] = t;
    }
    __syncthreads();
    const bf16* __restrict__ Bg = W + (size_t)e * 2 * (size_t)N * K;
    const bf16* aSrc[GPW_A];
    const bf16* bSrc[GPW_B];
#pragma unroll
    for (int i = 0; i < GPW_A; ++i) {
      const int row = (wave * GPW_A + i) * 8 + grow8;
      const size_t arow = (size_t)(sTps[row].tokenIdx & 0x0FFFFFFF);
      aSrc[i] = x + arow * K + schunk * 8;
    }
#pragma unroll
    for (int i = 0; i < GPW_B; ++i) {
      const int row = (wave * GPW_B + i) * 8 + grow8;
      bSrc[i] = Bg + (size_t)min(n0 + row, N - 1) * K + schunk * 8;
    }
    auto stage = [&](int kt, int buf) {
#pragma unroll
      for (int i = 0; i < GPW_A; ++i)
        __builtin_amdgcn_global_load_lds(
            (gas_u32p*)(aSrc[i] + kt),
            (las_u32p*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 512), 16, 0, 0);
#pragma unroll
      for (int i = 0; i < GPW_B; ++i)
        __builtin_amdgcn_global_load_lds(
            (gas_u32p*)(bSrc[i] + kt),
            (las_u32p*)(Bbase + buf * BN * BK + (wave * GPW_B + i) * 512), 16, 0, 0);
    };
    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    for (int t = 0; t < nK; ++t) {
      const bool stageNow = t + 1 < nK;
      const bool late = wave >= 4;
      if (stageNow && !late) stage((t + 1) * BK, (t + 1) & 1);
      const bf16* Al = Abase + (t & 1) * BM * BK;
      const bf16* Bl = Bbase + (t & 1) * BN * BK;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        if (ss == 1 && late && stageNow) {
          __builtin_amdgcn_s_setprio(0);
          stage((t + 1) * BK, (t + 1) & 1);
          __builtin_amdgcn_s_setprio(1);
        }
        bf16x8 af[8], bfr[4];
        const int cbase = 4 * ss + (lane >> 4);
        auto aread = [&](int mi) {
          const int R = wr * 128 + mi * 16 + rl;
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
        };
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int R = wc * 64 + ni * 16 + rl;
          bfr[ni] = *reinterpret_cast<const bf16x8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
        aread(0); aread(1);
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g2 = 0; g2 < 4; ++g2) {
          if (g2 < 3) { aread(2 * g2 + 2); aread(2 * g2 + 3); }
#pragma unroll
          for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
              acc[mi][ni] = MFMA(af[mi], bfr[ni], acc[mi][ni]);
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    const int cl = lane & 15;
    const int r0 = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = wr * 128 + mi * 16 + r0 + r;
        const int m = m0 + row;
        if ((uint32_t)m >= routed) continue;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int col = n0 + wc * 64 + ni * 16 + cl;
          if (col >= N) continue;
          float v = acc[mi][ni][r];
          v = fmaxf(v, 0.0f);
          xM[(size_t)e * pEC * N + (size_t)m * N + col] = (bf16)v;
        }
      }
    }
  }
}

static double run(void (*kern)(const bf16*, float*, int), const bf16* seed,
                  float* sink, int blocks, int iters) {
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, seed, sink, iters);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(e0);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, seed, sink, iters);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  // flops: blocks x 8 waves x iters x 2 s x 32 mfma x 16384 flops
  const double fl = (double)blocks * 8 * iters * 2 * 32 * 16384;
  return fl / (ms * 1e-3) / 1e12;
}

static double run_stage(void (*kern)(const bf16*, const bf16*, float*, int, int),
                        const bf16* A, const bf16* B, float* sink, int blocks,
                        int nK, int iters) {
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, A, B, sink, nK, iters);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(e0);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, A, B, sink, nK, iters);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  // NF=2 (BN=128) flops; the BN=256 caller doubles the printed figure
  const double fl = (double)blocks * 8 * iters * (double)nK * 2 * 16 * 16384;
  return fl / (ms * 1e-3) / 1e12;
}

int main() {
  bf16* seed;
  float* sink;
  (void)hipMalloc(&seed, 1 << 20);
  (void)hipMalloc(&sink, 1 << 20);
  (void)hipMemset(seed, 0x3c, 1 << 20);
  const int iters = 20000;
  for (int blocks : {256, 512}) {
    double t1 = run(k_mfma_max, seed, sink, blocks, iters);
    double t2 = run(k_lds_mfma<false>, seed, sink, blocks, iters);
    double t3 = run(k_lds_mfma<true>, seed, sink, blocks, iters);
    printf("blocks=%d  mfma_max %.0f TF  lds_plain %.0f TF  lds_asm %.0f TF\n",
           blocks, t1, t2, t3);
  }
  // staged variants: A 512 rows x K, B 8 panels of 256 x K (mimics the
  // up-GEMM working set: A streamed once, B panels L2-hot per XCD)
  const int nK = 16;  // K = 1024 (cfg2 up)
  const int K = nK * 64;
  bf16 *A, *B;
  (void)hipMalloc(&A, (size_t)512 * 256 * K * 2);   // >= blocks*BM rows
  (void)hipMalloc(&B, (size_t)64 * 256 * K * 2);
  (void)hipMemset(A, 0x3c, (size_t)512 * 256 * K * 2);
  (void)hipMemset(B, 0x3c, (size_t)64 * 256 * K * 2);
  const int sIters = 400;
  for (int blocks : {256, 512}) {
    double g2 = run_stage(k_stage_gemm<2, false, 256>, A, B, sink, blocks, nK, sIters);
    printf("blocks=%d nK=%d  staged2-BN256 %.0f TF (of 2104 LDS-resident)\n",
           blocks, nK, g2 * 2);  // NF=4: formula below assumes NF=2
    double h2 = run_stage(k_stage_gemm<2, false, 128>, A, B, sink, blocks, nK, sIters);
    double h3 = run_stage(k_stage_gemm<3, false, 128>, A, B, sink, blocks, nK, sIters);
    double h3c = run_stage(k_stage_gemm<3, true, 128>, A, B, sink, blocks, nK, sIters);
    printf("blocks=%d nK=%d  staged2-BN128 %.0f TF  staged3-drain %.0f TF  "
           "staged3-counted %.0f TF\n", blocks, nK, h2, h3, h3c);
    double sc2 = run_stage(k_stage_gemm<2, false, 256, true>, A, B, sink,
                           blocks, nK, sIters);
    double sc3 = run_stage(k_stage_gemm<3, true, 128, true>, A, B, sink,
                           blocks, nK, sIters);
    printf("blocks=%d nK=%d  SCATTERED-A: staged2-BN256 %.0f TF  "
           "staged3c-BN128 %.0f TF\n", blocks, nK, sc2 * 2, sc3);
    double bb = run_stage(k_stage_gemm<2, false, 256, true, 1>, A, B, sink,
                          blocks, nK, sIters);
    double be = run_stage(k_stage_gemm<2, false, 256, true, 1, true>, A, B,
                          sink, blocks, nK, sIters);
    printf("blocks=%d nK=%d  +HBM-B(32MB) %.0f TF  +epilogue %.0f TF\n",
           blocks, nK, bb * 2, be * 2);
    double cb = run_stage(k_stage_gemm<2, false, 256, true, 2, true>, A, B,
                          sink, blocks, nK, sIters);
    double cb3 = run_stage(k_stage_gemm<3, true, 128, true, 2, true>, A, B,
                           sink, blocks, nK, sIters);
    printf("blocks=%d nK=%d  COLD-B: staged2-BN256 %.0f TF  "
           "staged3c-BN128 %.0f TF\n", blocks, nK, cb * 2, cb3);
  }
  // variant 5: full real structure. cfg2 up shapes: K=1024, N=4096,
  // pEC=1024, E=8; grid (4,16,8); tokenIds scattered in a 4096-row x.
  {
    const int K5 = 1024, N5 = 4096, pEC5 = 1024, E5 = 8;
    bf16 *x5, *W5, *xM5;
    MiniTPS* tps5;
    uint32_t* eC5;
    (void)hipMalloc(&x5, (size_t)4096 * K5 * 2);
    (void)hipMalloc(&W5, (size_t)E5 * 2 * N5 * K5 * 2);  // 128 MB
    (void)hipMalloc(&xM5, (size_t)E5 * pEC5 * N5 * 2);
    (void)hipMalloc(&tps5, (size_t)E5 * pEC5 * sizeof(MiniTPS));
    (void)hipMalloc(&eC5, E5 * sizeof(uint32_t));
    (void)hipMemset(x5, 0x3c, (size_t)4096 * K5 * 2);
    (void)hipMemset(W5, 0x3c, (size_t)E5 * 2 * N5 * K5 * 2);
    MiniTPS* htps = new MiniTPS[E5 * pEC5];
    for (int e = 0; e < E5; ++e)
      for (int i = 0; i < pEC5; ++i)
        htps[e * pEC5 + i] = MiniTPS{(uint32_t)((e * pEC5 + i * 997) & 4095), 1.0f};
    uint32_t hec[E5];
    for (int e = 0; e < E5; ++e) hec[e] = pEC5;
    (void)hipMemcpy(tps5, htps, (size_t)E5 * pEC5 * sizeof(MiniTPS),
                    hipMemcpyHostToDevice);
    (void)hipMemcpy(eC5, hec, sizeof(hec), hipMemcpyHostToDevice);
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    const int it5 = 200;
    dim3 g5(pEC5 / 256, N5 / 256, E5);
    hipLaunchKernelGGL(k_stage_real, g5, dim3(512), 0, 0, x5, W5, xM5, tps5,
                       eC5, K5, N5, pEC5, 4);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(k_stage_real, g5, dim3(512), 0, 0, x5, W5, xM5, tps5,
                       eC5, K5, N5, pEC5, it5);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    const double fl = 2.0 * E5 * pEC5 * (double)N5 * K5 * it5;
    printf("REAL-STRUCTURE cfg2-up amortized(it=200): %.0f TF (%.1f us per fwd-equiv)\n",
           fl / (ms * 1e-3) / 1e12, ms * 1e3 / it5);
    // per-launch variant: iters=1, 50 back-to-back launches - isolates
    // the per-launch cost the persistent grid was meant to amortize
    (void)hipEventRecord(e0);
    for (int r = 0; r < 50; ++r)
      hipLaunchKernelGGL(k_stage_real, g5, dim3(512), 0, 0, x5, W5, xM5, tps5,
                         eC5, K5, N5, pEC5, 1);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    (void)hipEventElapsedTime(&ms, e0, e1);
    const double fl1 = 2.0 * E5 * pEC5 * (double)N5 * K5 * 50;
    printf("REAL-STRUCTURE cfg2-up per-launch(it=1 x50): %.0f TF (%.1f us per launch)\n",
           fl1 / (ms * 1e-3) / 1e12, ms * 1e3 / 50);

    // queue-driven variant: same tiles via task descriptors + atomic claim
    const int J = (pEC5 / 256) * (N5 / 256) * E5;  // 512 tiles
    TileTask* tasks;
    unsigned long long* claim;
    (void)hipMalloc(&tasks, (size_t)J * sizeof(TileTask));
    (void)hipMalloc(&claim, sizeof(unsigned long long));
    {
      TileTask* ht = new TileTask[J];
      const int mT = pEC5 / 256, nT = N5 / 256;
      for (int lin = 0; lin < J; ++lin) {
        // same XCD remap order as the blockIdx decomposition
        const int qx = J / 8, rx = J % 8;
        const int xcd = lin % 8, pos = lin / 8;
        const int swz =
            (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
        const int e = swz / (mT * nT), rem = swz % (mT * nT);
        ht[lin] = TileTask{(uint32_t)e, (uint32_t)((rem % mT) * 256),
                           (uint32_t)((rem / mT) * 256), 0};
      }
      (void)hipMemcpy(tasks, ht, (size_t)J * sizeof(TileTask),
                      hipMemcpyHostToDevice);
      delete[] ht;
    }
    const int itq = 200;
    (void)hipMemset(claim, 0, sizeof(unsigned long long));
    hipLaunchKernelGGL(k_stage_queued, dim3(512), dim3(512), 0, 0, x5, W5,
                       xM5, tps5, eC5, tasks, claim, J * 4, J, K5, N5, pEC5);
    (void)hipDeviceSynchronize();
    (void)hipMemset(claim, 0, sizeof(unsigned long long));
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(k_stage_queued, dim3(512), dim3(512), 0, 0, x5, W5,
                       xM5, tps5, eC5, tasks, claim, J * itq, J, K5, N5, pEC5);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    (void)hipEventElapsedTime(&ms, e0, e1);
    const double flq = 2.0 * E5 * pEC5 * (double)N5 * K5 * itq;
    printf("QUEUE-DRIVEN cfg2-up: %.0f TF (%.1f us per forward-equiv) - "
           "task-dispatch overhead vs REAL-STRUCTURE\n",
           flq / (ms * 1e-3) / 1e12, ms * 1e3 / itq);

    // dynamic pipeline: 32 producers (simulated ~15 us route delay) + 480
    // consumers, single launch per forward; compare wall vs 15 us + GEMM
    uint32_t* seqp;
    TileTask* ringp;
    (void)hipMalloc(&seqp, (size_t)J * sizeof(uint32_t));
    (void)hipMalloc(&ringp, (size_t)J * sizeof(TileTask));
    // delay loop is scratch-bound (~100 cyc/iter): 0 / ~20 us / ~60 us
    for (int delay : {0, 500, 5000, 15000}) {
      float tot = 0;
      const int reps = 30;
      for (int r = 0; r < reps; ++r) {
        (void)hipMemset(seqp, 0, (size_t)J * sizeof(uint32_t));
        (void)hipMemset(claim, 0, sizeof(unsigned long long));
        (void)hipEventRecord(e0);
        hipLaunchKernelGGL(k_stage_pipe, dim3(512), dim3(512), 0, 0, x5, W5,
                           xM5, tps5, eC5, tasks, ringp, seqp, claim, 32, J,
                           delay, K5, N5, pEC5);
        (void)hipEventRecord(e1);
        (void)hipEventSynchronize(e1);
        (void)hipEventElapsedTime(&ms, e0, e1);
        tot += ms;
      }
      printf("PIPELINE delayLoops=%d: %.1f us per forward "
             "(480 consumers; producer delay overlapped)\n",
             delay, tot / reps * 1e3);
    }
  }
  return 0;
}
