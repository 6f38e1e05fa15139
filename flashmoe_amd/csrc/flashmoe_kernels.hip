// MI355X-native (gfx950 / CDNA4) implementation of the FlashDMoE hot path:
//   gate -> token routing -> grouped expert FFN (MFMA) -> weighted combine
// Built from scratch against the C-ABI in include/flashmoe_abi.h; semantics
// restated from the reference (citations per kernel, file:line into
// /root/reference). No CUDA compatibility paths; wave64 / MFMA / LDS only.
//
// Round-1 structure: separate kernels on one stream.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <algorithm>

#include "../../include/flashmoe_abi.h"

// ---------------------------------------------------------------------------
// Common types / helpers
// ---------------------------------------------------------------------------

using bf16 = __hip_bfloat16;

// token slot descriptor (reference TPS, types.cuh:299-312): token index +
// sum of the token's selected top-k probabilities (gate.cuh:669,711-715).
// The top 4 bits of tokenIdx carry the assignment's position j within the
// token's top-k (tokens <= 2^28; lets the combine write non-atomic
// per-(token, j) slots instead of fp32 atomics).
struct __align__(8) TPS {
  uint32_t tokenIdx;
  float probSum;
};
__device__ __host__ __forceinline__ uint32_t tpsTok(uint32_t v) {
  return v & 0x0FFFFFFFu;
}
__device__ __host__ __forceinline__ uint32_t tpsJ(uint32_t v) { return v >> 28; }

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) _Float16 halfx8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;

using fp16 = __half;
struct fp8e4m3 { uint8_t v; };  // storage-only; dequant at fragment read

// counted vmcnt wait with a compile-time literal (the triple-buffered
// staging pipeline waits for tile t+1 while t+2's glds stay in flight)
template <int N> __device__ __forceinline__ void wait_vmcnt() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 1) asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else if constexpr (N == 3) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 10) asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
  else if constexpr (N == 12) asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt literal");
}

typedef float f32x2n __attribute__((ext_vector_type(2)));
// convert 8 packed fp8-e4m3 (as 2 dwords) to 8 bf16 lanes - exact: every
// e4m3 value is bf16-representable (used at B fragment read, dtype 4)
__device__ __forceinline__ bf16x8 dequant_fp8x8_bf16(uint32_t lo, uint32_t hi) {
  bf16x8 r;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const uint32_t w = p ? hi : lo;
    const f32x2n f0 = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);  // bytes 0,1
    const f32x2n f1 = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);   // bytes 2,3
    r[p * 4 + 0] = (__bf16)__float2bfloat16(f0.x);
    r[p * 4 + 1] = (__bf16)__float2bfloat16(f0.y);
    r[p * 4 + 2] = (__bf16)__float2bfloat16(f1.x);
    r[p * 4 + 3] = (__bf16)__float2bfloat16(f1.y);
  }
  return r;
}

// 2-byte element store, optionally agent-scope write-through (sc1) so
// same-launch consumers on other CUs/XCDs observe it without the
// producer issuing a cache-flushing release fence (Guideline 16 R1)
template <typename ET>
__device__ __forceinline__ void storeElem(ET* p, ET v, bool sc1) {
  if (sc1) {
    uint16_t b;
    __builtin_memcpy(&b, &v, 2);
    __hip_atomic_store(reinterpret_cast<uint16_t*>(p), b, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
  } else {
    *p = v;
  }
}

__device__ __forceinline__ float toF(float v) { return v; }
__device__ __forceinline__ float toF(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ float toF(fp16 v) { return __half2float(v); }
__device__ __forceinline__ void fromF(float v, float& o) { o = v; }
__device__ __forceinline__ void fromF(float v, bf16& o) { o = __float2bfloat16(v); }
__device__ __forceinline__ void fromF(float v, fp16& o) { o = __float2half(v); }

// element traits for the low-precision MFMA GEMM kernels (bf16 / fp16;
// both K=32 16x16 MFMA shapes share operand/result layouts on gfx950)
template <typename ET> struct ETr;
template <> struct ETr<bf16> {
  using vec8 = bf16x8;
  static __device__ __forceinline__ f32x4 mfma(vec8 a, vec8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ bf16 fromf(float v) { return __float2bfloat16(v); }
};
template <> struct ETr<fp16> {
  using vec8 = halfx8;
  static __device__ __forceinline__ f32x4 mfma(vec8 a, vec8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
  static __device__ __forceinline__ fp16 fromf(float v) { return __float2half(v); }
};

// hidden_act 0: ReLU, 1: exact-erf GELU (cutlass epilogue::thread::{ReLU,
// GELU}, schema csrc/flashmoe_config.schema.json:33-37, types.cuh:151-159)
// - applied inline in the GEMM epilogues via the compile-time ACT param.

#define DIVUP(a, b) (((a) + (b) - 1) / (b))

// ---------------------------------------------------------------------------
// Split gate (v2): the single-kernel gate ran on only S/128 blocks (32 for
// config 2 - 12% of the CUs, 61% wave-park, profiles/r01). k_gate_logits
// parallelises the logit GEMM over (token-tile, H-chunk) blocks with fp32
// atomicAdd partials into a global logits buffer; k_gate_route then does
// softmax/top-k/ordering per token tile (same semantics as k_gate,
// gate.cuh:474-720).
// ---------------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(256) void k_gate_logits(
    const T* __restrict__ x, const T* __restrict__ gate_w,
    float* __restrict__ logits32, uint32_t* __restrict__ eC, int S, int H,
    int E, int Hc) {
  // fold the eC zeroing into the first block (route runs after this
  // kernel; saves a 4-us launch-bound memset per forward)
  if (blockIdx.x == 0 && blockIdx.y == 0 && blockIdx.z == 0) {
    for (int i = threadIdx.x; i < E; i += blockDim.x) eC[i] = 0;
  }
  // blockIdx.z selects a chunk of <=128 experts (E up to 256, config 5)
  const int eBase = blockIdx.z * 128;
  const int Ec = min(E - eBase, 128);
  constexpr int BM = 128;
  constexpr int BK = 64;
  constexpr int RPAD = 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* part = reinterpret_cast<float*>(smem);          // [BM][Ec+1]
  T* aCh = nullptr;  // assigned after Ec is known (see below)
  const int tid = threadIdx.x;
  const int m0 = blockIdx.x * BM;
  const int tok = tid & (BM - 1);
  const int half = tid >> 7;
  const int E2 = (Ec + 1) / 2;
  const int e0 = half * E2;
  const int e1 = min(e0 + E2, Ec);
  const int k0 = blockIdx.y * Hc;
  const int k1 = min(H, k0 + Hc);
  aCh = reinterpret_cast<T*>(part + BM * (Ec + 1));      // [BM][BK+RPAD]
  const int LDA = BK + RPAD;
  T* gCh = aCh + BM * LDA;                               // [Ec][BK+RPAD]

  for (int i = tid; i < BM * (Ec + 1); i += 256) part[i] = 0.0f;
  for (int kc = k0; kc < k1; kc += BK) {
    constexpr int EPU = 16 / sizeof(T);
    {
      const int unitsA = BM * BK / EPU;
      for (int u = tid; u < unitsA; u += 256) {
        const int row = u / (BK / EPU);
        const int cu = u % (BK / EPU);
        *reinterpret_cast<u32x4*>(aCh + row * LDA + cu * EPU) =
            *reinterpret_cast<const u32x4*>(x + (size_t)(m0 + row) * H + kc + cu * EPU);
      }
      const int unitsG = Ec * BK / EPU;
      for (int u = tid; u < unitsG; u += 256) {
        const int row = u / (BK / EPU);
        const int cu = u % (BK / EPU);
        *reinterpret_cast<u32x4*>(gCh + row * LDA + cu * EPU) =
            *reinterpret_cast<const u32x4*>(gate_w + (size_t)(eBase + row) * H +
                                            kc + cu * EPU);
      }
    }
    __syncthreads();
    const T* arow = aCh + tok * LDA;
    u32x4 a8[BK * sizeof(T) / 16];
#pragma unroll
    for (int jb = 0; jb < BK * (int)sizeof(T) / 16; ++jb)
      a8[jb] = *reinterpret_cast<const u32x4*>(&arow[jb * (16 / sizeof(T))]);
    for (int e = e0; e < e1; ++e) {
      const T* grow = gCh + e * LDA;
      float sAcc = 0.0f;
#pragma unroll
      for (int jb = 0; jb < BK * (int)sizeof(T) / 16; ++jb) {
        const u32x4 g8 = *reinterpret_cast<const u32x4*>(&grow[jb * (16 / sizeof(T))]);
#pragma unroll
        for (int w = 0; w < 4; ++w) {
          const uint32_t aw = a8[jb][w], gw = g8[w];
          if constexpr (__is_same(T, bf16)) {
            const float2 av = __bfloat1622float2(
                *reinterpret_cast<const __hip_bfloat162*>(&aw));
            const float2 gv = __bfloat1622float2(
                *reinterpret_cast<const __hip_bfloat162*>(&gw));
            sAcc = fmaf(av.x, gv.x, sAcc);
            sAcc = fmaf(av.y, gv.y, sAcc);
          } else if constexpr (__is_same(T, fp16)) {
            const float2 av = __half22float2(*reinterpret_cast<const __half2*>(&aw));
            const float2 gv = __half22float2(*reinterpret_cast<const __half2*>(&gw));
            sAcc = fmaf(av.x, gv.x, sAcc);
            sAcc = fmaf(av.y, gv.y, sAcc);
          } else {
            sAcc = fmaf(__uint_as_float(aw), __uint_as_float(gw), sAcc);
          }
        }
      }
      part[tok * (Ec + 1) + e] += sAcc;
    }
    __syncthreads();
  }
  // one atomic per (token, expert) per H-chunk
  for (int e = e0; e < e1; ++e)
    atomicAdd(logits32 + (size_t)(m0 + tok) * E + eBase + e,
              part[tok * (Ec + 1) + e]);
}

// ---------------------------------------------------------------------------
// Softmax / top-k / ordered placement for one 128-token tile (reference
// gate.cuh:557-718 semantics): the shared core of k_gate_route (classic
// path) and the fused kernel's inline route. 512 threads; the per-token
// top-k runs on FOUR threads per token, each owning a <=64-expert chunk
// with a u64 taken-mask and combining candidates with width-4 shuffles
// (larger value wins, equal value -> smaller index == the reference's
// ascending strict-> first-index scan). The serial one-thread-per-token
// scan was the dominant gate cost at E=256 (~144 us at the cfg5 shape).
// Reads the tile's logits from logits32 and RE-ZEROES them (keeps the
// split-K logits accumulate memset-free).
// ---------------------------------------------------------------------------
template <typename T, int K>
__device__ __forceinline__ void route_tile_core(
    char* smem, float* __restrict__ logits32, T* __restrict__ gate_out,
    TPS* __restrict__ tokenIds, uint32_t* __restrict__ eC,
    uint8_t* __restrict__ kept, float* __restrict__ gML,
    float* __restrict__ gMeC, int S, int E, int PX, int EC, int pEC,
    int m0) {
  // TT tokens per pass: 128 normally; 64 half-passes at E > 256 so the
  // fp32 logits tile fits LDS (64 x 513 x 4 B = 131 KB at E = 512).
  // SUBS = 512/TT threads cooperate per token (chunk <= 64 experts each,
  // so the taken-mask stays one u64). Pass order preserves token order,
  // so intra-tile placement semantics are unchanged.
  const int TT = (E > 256) ? 64 : 128;
  const int SUBS = 512 / TT;
  float* logits = reinterpret_cast<float*>(smem);                  // [TT][E+1]
  uint16_t* sel = reinterpret_cast<uint16_t*>(logits + TT * (E + 1));
  uint16_t* localIdx = sel + TT * K;
  // +2: one dump slot so the counting scan's store is UNCONDITIONAL
  // (an exec-masked conditional store compiled into ~520 serial
  // s_and_saveexec blocks = ~80% of the round-1 kernel's runtime)
  uint32_t* base = reinterpret_cast<uint32_t*>(localIdx + TT * K + 2);  // [E]
  float* sInv = reinterpret_cast<float*>(base + E);                // [TT] 1/d
  float* sMax = sInv + TT;                                         // [TT]
  float* sCw = sMax + TT;                                          // [TT] mCw
  const int tid = threadIdx.x;
  for (int pass = 0; pass < 128 / TT; ++pass) {
  const int pm0 = m0 + pass * TT;
  __syncthreads();  // prior pass's LDS readers done
  for (int i = tid; i < TT * E; i += 512) {
    const int row = i / E, col = i % E;
    float* src = &logits32[(size_t)(pm0 + row) * E + col];
    logits[row * (E + 1) + col] = *src;
    *src = 0.0f;
  }
  __syncthreads();

  {
    const int token = tid / SUBS;
    const int sub = tid % SUBS;
    if (token < TT) {
      const float* lrow = logits + token * (E + 1);
      const int chunk = (E + SUBS - 1) / SUBS;  // <= 64: u64 taken-mask
      const int e0 = sub * chunk;
      const int e1 = min(E, e0 + chunk);
      float m = -INFINITY;
      for (int e = e0; e < e1; ++e) m = fmaxf(m, lrow[e]);
#pragma unroll
      for (int off = 1; off < 8; off <<= 1)
        if (off < SUBS) m = fmaxf(m, __shfl_xor(m, off, SUBS));
      float d = 0.0f;
      for (int e = e0; e < e1; ++e) d += __expf(lrow[e] - m);
#pragma unroll
      for (int off = 1; off < 8; off <<= 1)
        if (off < SUBS) d += __shfl_xor(d, off, SUBS);
      const float inv_d = 1.0f / d;
      if (sub == 0) {
        sInv[token] = inv_d;  // the coalesced gate_out pass below
        sMax[token] = m;      // recomputes probs from these
      }
      unsigned long long takenM = 0ull;
      float mCw = 0.0f;
#pragma unroll
      for (int i = 0; i < K; ++i) {
        float lv = -INFINITY;
        int li = E;  // sentinel sorts after every real index
        for (int e = e0; e < e1; ++e) {
          const bool tk = (takenM >> (e - e0)) & 1ull;
          if (!tk && lrow[e] > lv) { lv = lrow[e]; li = e; }
        }
#pragma unroll
        for (int off = 1; off < 8; off <<= 1) {
          if (off < SUBS) {
            const float ov = __shfl_xor(lv, off, SUBS);
            const int oi = __shfl_xor(li, off, SUBS);
            if (ov > lv || (ov == lv && oi < li)) { lv = ov; li = oi; }
          }
        }
        if (li >= e0 && li < e1) takenM |= 1ull << (li - e0);
        if (sub == 0) {
          sel[token * K + i] = (uint16_t)li;
          mCw += __expf(lv - m) * inv_d;
        }
      }
      if (sub == 0) sCw[token] = mCw;
    }
  }
  __syncthreads();
  // cooperative COALESCED gate_out store (recompute probs from LDS; a
  // per-token scalar store pattern was one cache line per element)
  for (int i = tid * 8; i < TT * PX; i += 512 * 8) {
    const int row = i / PX, col0 = i % PX;
    const float mR = sMax[row], ivR = sInv[row];
    const float* lrow = logits + row * (E + 1);
    struct __attribute__((aligned(16))) V8 { T x[8]; } v;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      const int col = col0 + q;
      const float pv = (col < E) ? __expf(lrow[col] - mR) * ivR : 0.0f;
      fromF(pv, v.x[q]);
    }
    *reinterpret_cast<V8*>(gate_out + (size_t)(pm0 + row) * PX + col0) = v;
  }
  __syncthreads();
  if (tid < E) {
    // 16-B LDS reads, 8 selections per iteration (a scalar dependent
    // scan exposed ~50 cycles of LDS latency per entry: ~7 us/fwd)
    uint32_t cnt = 0;
    for (int mj = 0; mj < TT * K; mj += 8) {
      const u32x4 v = *reinterpret_cast<const u32x4*>(sel + mj);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const uint32_t w = v[q >> 1];
        const uint16_t sv = (q & 1) ? (uint16_t)(w >> 16) : (uint16_t)(w & 0xffff);
        const bool mt = (sv == (uint16_t)tid);
        localIdx[mt ? (mj + q) : TT * K] = (uint16_t)cnt;
        cnt += mt;
      }
    }
    base[tid] = __hip_atomic_fetch_add(eC + tid, cnt, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
    if (gML) {
      // column sum of probs for this expert over the pass's tokens
      float colSum = 0.0f;
      for (int mj = 0; mj < TT; ++mj)
        colSum += __expf(logits[mj * (E + 1) + tid] - sMax[mj]) * sInv[mj];
      atomicAdd(gML + tid, colSum / (float)S);
      atomicAdd(gMeC + tid, (float)cnt / (float)S);
    }
  }
  __syncthreads();
  if (tid < TT) {
#pragma unroll
    for (int i = 0; i < K; ++i) {
      const int e = sel[tid * K + i];
      const uint32_t slot = base[e] + localIdx[tid * K + i];
      const bool keep = slot < (uint32_t)EC;
      if (keep) {
        tokenIds[(size_t)e * pEC + slot] =
            TPS{(uint32_t)(pm0 + tid) | ((uint32_t)i << 28), sCw[tid]};
      }
      if (kept) kept[(size_t)(pm0 + tid) * K + i] = keep ? 1 : 0;
    }
  }
  }  // pass loop
}

template <typename T, int K>
__global__ __launch_bounds__(512) void k_gate_route(
    float* __restrict__ logits32, T* __restrict__ gate_out,
    TPS* __restrict__ tokenIds, uint32_t* __restrict__ eC, int S, int E,
    int PX, int EC, int pEC, float* __restrict__ gML,
    float* __restrict__ gMeC, uint8_t* __restrict__ kept) {
  // gML/gMeC: training-mode aux-loss accumulators (gate.cuh:273-299,
  // 763-773; types.cuh:936-958). Null in inference mode.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  route_tile_core<T, K>(smem, logits32, gate_out, tokenIds, eC, kept, gML,
                        gMeC, S, E, PX, EC, pEC, blockIdx.x * 128);
}

// ---------------------------------------------------------------------------
// MX-fp8 activation quantization (dtype 5, the cfg5 regime: "CDNA4 fp8
// MFMA + token-scaling"): bf16 rows -> fp8 e4m3 bytes + per-64-element
// E8M0 block scales. Block size 64 matches the EMPIRICAL scale
// granularity of v_mfma_scale_f32_16x16x128_f8f6f4 on gfx950 (probe
// test_mx_mfma_layout_probe: the scale byte of lane-group g applies to
// the interleaved chunk pair {k: k>>6 == g&1, (k>>4)&1 == g>>1}, so a
// per-contiguous-64 scale supplied to lane-groups {0,2} (block 0) and
// {1,3} (block 1) scales every hardware block uniformly; opsel selects
// the byte). Scale = 2^e with e chosen so blockmax/2^e <= 448 (e4m3
// max); rounding RNE via v_cvt_pk_fp8_f32.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint32_t cvt2_fp8(float a, float b) {
  // pack two fp32 -> two e4m3 bytes (RNE), low half of a dword
  f32x2n v;
  v.x = a;
  v.y = b;
  return __builtin_amdgcn_cvt_pk_fp8_f32(v.x, v.y, 0u, false);
}

template <typename T>
__global__ void k_quant_mx(const T* __restrict__ in, uint8_t* __restrict__ out,
                           uint8_t* __restrict__ scales, long long nRows,
                           int K) {
  // one wave per 64-element block: lane l holds in[row][b*64 + l] ...
  // simpler: each thread walks one block serially is too slow; use
  // 64 lanes x 1 element? Layout: block j of row r -> 64 values.
  // Grid-stride over blocks; blockDim 256 = 4 waves, each wave one
  // 64-block per iteration (lane = element).
  const int wave = (int)(threadIdx.x >> 6);
  const int lane = (int)(threadIdx.x & 63);
  const long long nBlk = nRows * (K / 64);
  for (long long b = (long long)blockIdx.x * 4 + wave; b < nBlk;
       b += (long long)gridDim.x * 4) {
    const long long row = b / (K / 64);
    const int j = (int)(b % (K / 64));
    const float v = toF(in[row * K + j * 64 + lane]);
    // wave max |v|
    float m = fabsf(v);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      m = fmaxf(m, __shfl_xor(m, off, 64));
    // E8M0 exponent e: smallest power of two with blockmax/2^e <= 448
    int e = 0;
    if (m > 0.0f) {
      int ex;
      (void)frexpf(m / 448.0f, &ex);  // m/448 = f * 2^ex, f in [0.5, 1)
      e = ex;                          // 2^ex >= m/448 > 2^(ex-1)
      if (e < -126) e = -126;
      if (e > 127) e = 127;
    }
    const float inv = exp2f((float)-e);
    const uint32_t pk = cvt2_fp8(v * inv, 0.0f);
    out[row * K + j * 64 + lane] = (uint8_t)(pk & 0xFF);
    if (lane == 0) scales[row * (K / 64) + j] = (uint8_t)(127 + e);
  }
}

// ---------------------------------------------------------------------------
// Grouped expert GEMM (reference: os/processor/processor.cuh fGET preGEMM
// :339-468 / postGEMM :711-751, gemm.cuh FAA epilogue, combine :44-205).
// PHASE 0 (up):   A = x rows gathered via tokenIds[e]; B = Wup[e] [P,H];
//                 epilogue act(acc + b_up) -> xM[e]
// PHASE 1 (down): A = xM[e]; B = Wdn[e] viewed [H,P] (reference
//                 reinterpretation, moe.cuh:114-116); epilogue
//                 z = acc + b_dn; k>1: atomicAdd(O32, gate_out*z/probSum)
//                 (processor.cuh:126-168); k==1: unscaled store to moe_out
//                 (processor.cuh:173-204)
// PHASE 2 (down-direct, EP path): like 1 but plain store to out_rows[m].
// PHASE 3 (gate logits): A = x (identity rows), B = gate_w viewed [E,H]
//                 (reference quirk, moe.cuh:107-109); epilogue: plain f32
//                 store to out[m*N + col] - feeds k_gate_route. MFMA
//                 replaces the VALU logits kernel (it was ~80 us at the
//                 E=64/H=1024 8-GPU shape).
// If tokenIds == nullptr the row gather is identity and routed = n_rows
// (packed-rows mode for fm_expert_ffn).
// ---------------------------------------------------------------------------

struct GemmArgs {
  const void* A;        // phase0: x [S,H]; phase1/2: xM_e rows [pEC,K]
  const void* B;        // weight rows [N,K] (K contiguous)
  const void* bias;     // [N] or null
  void* out;            // phase0: xM_e [pEC,P]; phase2: out_rows
  float* O32;           // phase1 k>1
  void* moe_out;        // phase1 k==1
  const void* gate_out; // [S,PX]
  const TPS* tokenIds;  // [E,pEC] (this expert's row = tokenIds + e*pEC)
  const uint32_t* eC;
  long long strideAExpert;  // elements between experts in A (phase1: pEC*K)
  long long strideBExpert;  // elements between experts in B (2*P*H)
  long long strideOExpert;  // elements between experts in out
  int K;                // reduction dim
  int N;                // output dim
  int EC, pEC, PX;
  int topk;
  int act;
  int expertOffset;     // global expert id of blockIdx.z==0
  int nRows;            // packed-rows mode row count
  int H;                // row stride of x / O32 / moe_out
  int splitK;           // K-split factor (PHASE 1 multi only: the fp32
                        // atomicAdd combine makes split-K partials free)
  const int32_t* segExpert;  // optional [gridZ] device map: z -> weight
                             // expert (padded-EP segments; null = identity)
  int noRemap;               // debug: 1 disables the XCD block remap
  int totalJobs;             // persistent grid: total (m,n,e) tiles to
                             // cover (0 = one tile per block, classic)
  int jobsMT, jobsNT;        // tile grid dims when persistent
  int atomicLogits;          // PHASE 3: atomicAdd into fp32 out (K-split
                             // partial accumulate; fused gate)
  int slotAlways;            // PHASE 1: write the per-(token,j) combine
                             // slot even at topk==1 (scale 1; the fused
                             // kernel always reduces via k_cast_combine
                             // semantics so dropped tokens zero out)
  int sc1Out;                // epilogue stores agent-scope write-through
                             // (Guideline 16 R1 publish-large): in-launch
                             // consumers on other CUs see them after a
                             // vmcnt drain + job-count arrival, with no
                             // per-block release fence
  const void* aScales;       // MX path (dtype 5): per-64-element E8M0
                             // block scales of the quantized A operand,
                             // laid out like A with K/64 bytes per row
  void* out8;                // MX PHASE 0 (BN=256 geometry): quantize the
  void* outScales;           // activation tile in-register in the
                             // epilogue (bf16-round -> 16-lane blockmax
                             // -> e4m3) and store fp8 + E8M0 scales
                             // directly - skips the k_quant_mx pass and
                             // the bf16 xM round-trip entirely
};

// address-space helpers for global_load_lds (direct HBM->LDS DMA)
typedef __attribute__((address_space(1))) const uint32_t gas_u32;
typedef __attribute__((address_space(3))) uint32_t las_u32;

template <typename ET, int PHASE, int ACT, bool HAS_BIAS, typename WET = ET>
__global__ __launch_bounds__(256) void k_group_gemm_bf16(GemmArgs a) {
  using vec8 = typename ETr<ET>::vec8;
  constexpr int BM = 128, BN = 128, BK = 64;
  // WET is the WEIGHT (B operand) storage element: == ET normally, or
  // fp8e4m3 for the config-5 fp8-weight path (W8A16: B staged through
  // glds as raw fp8 bytes, dequantized to bf16 at fragment-read time so
  // the MFMA stream and the all-glds staging pipeline are unchanged and
  // B's HBM/LDS traffic halves).
  constexpr int BEZ = (int)sizeof(WET);
  constexpr int BCH = BK * BEZ / 16;  // 16B chunks per B row (8 or 4)
  // ONE shared arena (a second __shared__ object would force a vmcnt(0)
  // drain before every ds_read beside glds - guide par.5 trap 4a).
  // A/B tiles are LINEAR [128][64] bf16 (glds writes lane-linearly); the
  // bank swizzle lives on the SOURCE chunk index and the fragment-read
  // address (rule 21): chunk' = chunk ^ (row & 7) (bf16 B); for fp8 B
  // (64B rows) it is chunk16' = chunk16 ^ ((row >> 2) & 3), which makes
  // the b64 fragment reads bank-conflict-free.
  __shared__ __attribute__((aligned(16))) char smem[
      BM * BK * 2 + BN * BK * BEZ + BM * 8 + 16];
  ET* Alds = reinterpret_cast<ET*>(smem);
  WET* Blds = reinterpret_cast<WET*>(smem + BM * BK * 2);
  TPS* sTps = reinterpret_cast<TPS*>(smem + BM * BK * 2 + BN * BK * BEZ);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);

  const int skS = a.splitK > 0 ? a.splitK : 1;
  const int e = blockIdx.z % (gridDim.z / skS);
  const int ksplitS = blockIdx.z / (gridDim.z / skS);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int K = a.K, N = a.N;
  const int kLenS = K / skS;
  const int kStartS = ksplitS * kLenS;

  const int we = a.segExpert ? a.segExpert[e] : e;  // weight/bias expert
  if constexpr (PHASE == 3) {
    // gate-logits phase: block (0,0,0) zeroes eC for the route kernel
    // that follows (saves a launch-bound memset per forward); EC field
    // carries the expert count here
    if (a.eC && blockIdx.x == 0 && blockIdx.y == 0 && blockIdx.z == 0) {
      uint32_t* ec = const_cast<uint32_t*>(a.eC);
      for (int i = tid; i < a.EC; i += 256) ec[i] = 0;
    }
  }
  const TPS* tpsE = a.tokenIds ? a.tokenIds + (size_t)e * a.pEC : nullptr;
  if (tid == 0) {
    *sRouted = tpsE ? min(a.eC[e], (uint32_t)a.EC) : (uint32_t)a.nRows;
  }
  __syncthreads();
  const uint32_t routed = *sRouted;
  if ((uint32_t)m0 >= routed) return;  // empty tile (0-token expert / tail)

  if (tid < BM) {
    TPS t{0u, 1.0f};
    if ((uint32_t)(m0 + tid) < routed) {
      t = tpsE ? tpsE[m0 + tid] : TPS{(uint32_t)(m0 + tid), 1.0f};
    }
    sTps[tid] = t;
  }
  __syncthreads();

  const ET* __restrict__ Ag = reinterpret_cast<const ET*>(a.A);
  const WET* __restrict__ Bg =
      reinterpret_cast<const WET*>(a.B) + (size_t)we * a.strideBExpert;

  // accumulators: wave (wr,wc) owns the 64x64 subtile at (wr*64, wc*64)
  const int wr = wave >> 1, wc = wave & 1;
  f32x4 accv[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) accv[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // glds staging: 16 KiB per tile = 16 wave-instructions of 1 KiB; wave w
  // issues groups w*4..w*4+3; lane l covers row grp*8 + l/8, chunk l%8.
  // Source chunk is XOR-swizzled: sc = (l%8) ^ (l/8) (row&7 == l/8).
  const int grow8 = lane >> 3;           // row within the 8-row group
  const int schunk = (lane & 7) ^ grow8;  // swizzled 16B chunk index
  // B staging geometry in WET units: 1KiB glds group covers 1024/(BK*BEZ)
  // rows; lane -> (row-in-group, chunk) with the WET-specific swizzle.
  constexpr int BGW = BN * BK * BEZ / 1024 / 4;  // B groups per wave (4 or 2)
  const int bgrow = lane / BCH;
  const int bsc = (lane % BCH) ^ (BEZ == 1 ? ((bgrow >> 2) & 3) : bgrow);
  const int aRowStride = (PHASE == 0) ? a.H : K;
  const size_t aBase = (size_t)e * a.strideAExpert;  // 0 for the x-gather up phase
  for (int kt = kStartS; kt < kStartS + kLenS; kt += BK) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int grp = wave * 4 + i;
      const int row = grp * 8 + grow8;
      const size_t arow = (PHASE == 0) ? (size_t)tpsTok(sTps[row].tokenIdx)
                                       : (size_t)(m0 + row);
      const ET* asrc = Ag + aBase + arow * aRowStride + kt + schunk * 8;
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)asrc, (las_u32*)(Alds + grp * 512), 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < BGW; ++i) {
      const int grp = wave * BGW + i;
      const int row = grp * (1024 / (BK * BEZ)) + bgrow;
      const int brow = min(n0 + row, N - 1);
      const WET* bsrc = Bg + (size_t)brow * K + kt + bsc * (16 / BEZ);
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)bsrc, (las_u32*)((char*)Blds + grp * 1024), 16, 0, 0);
    }
    __syncthreads();  // carries the vmcnt(0) glds drain (guide par.5)
#pragma unroll
    for (int s = 0; s < 2; ++s) {  // two K=32 MFMA steps per tile
      vec8 af[4], bf[4];
      const int rl = lane & 15;
      const int cbase = 4 * s + (lane >> 4);  // 16B chunk before swizzle
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int R = wr * 64 + mi * 16 + rl;
        af[mi] = *reinterpret_cast<const vec8*>(
            &Alds[R * BK + ((cbase ^ (R & 7)) * 8)]);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int R = wc * 64 + ni * 16 + rl;
        if constexpr (BEZ == 1) {
          // fp8 B: 8-byte read (the (R>>2)&3 chunk16 swizzle makes the 16
          // lanes of each quarter-wave hit 16 distinct banks), dequant to
          // bf16 in regs before the MFMA
          const int c16 = (cbase >> 1) ^ ((R >> 2) & 3);
          const uint32_t* bp = reinterpret_cast<const uint32_t*>(
              (const char*)Blds + (size_t)R * BK + c16 * 16 + (cbase & 1) * 8);
          bf[ni] = dequant_fp8x8_bf16(bp[0], bp[1]);
        } else {
          bf[ni] = *reinterpret_cast<const vec8*>(
              &Blds[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          accv[mi][ni] = ETr<ET>::mfma(af[mi], bf[ni], accv[mi][ni]);
    }
    __syncthreads();
  }

  // epilogue: C/D map for 16x16 MFMA: col = lane&15, row = (lane>>4)*4 + r.
  // Bias and combine scales are HOISTED (a per-element bias load made
  // hipcc emit one dependent global_load+vmcnt(0) per output - guide
  // par.5 trap 4c, profiles/r01); activation is a compile-time template.
  const int cl = lane & 15;
  const int r0 = (lane >> 4) * 4;
  float* sScale = reinterpret_cast<float*>(Alds);  // reuse; loop is done
  const bool multi = (PHASE == 1) && a.topk > 1;
  if constexpr (PHASE == 1) {
    if (multi) {
      __syncthreads();
      if (tid < BM) {
        const TPS tp = sTps[tid];
        float sc = 0.0f;
        if ((uint32_t)(m0 + tid) < routed)
          sc = toF(reinterpret_cast<const ET*>(
                   a.gate_out)[(size_t)tpsTok(tp.tokenIdx) * a.PX +
                               a.expertOffset + e]) / tp.probSum;
        sScale[tid] = sc;
      }
      __syncthreads();
    }
  }
  float bv[4] = {0.f, 0.f, 0.f, 0.f};
  if constexpr (HAS_BIAS) {
    // multi-expert launches (strideBExpert > 0) carry per-expert bias
    // slabs [nLx, N]; the packed-rows single-expert API passes the
    // expert's own slab (stride 0)
    const ET* bptr = reinterpret_cast<const ET*>(a.bias) +
                     (a.strideBExpert ? (size_t)we * N : 0);
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wc * 64 + ni * 16 + cl;
      if (col < N) bv[ni] = toF(bptr[col]);
    }
  }
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wr * 64 + mi * 16 + r0 + r;
      const int m = m0 + row;
      if ((uint32_t)m >= routed) continue;
      const TPS tp = sTps[row];
      const float rowScale = multi ? sScale[row] : 1.0f;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int col = n0 + wc * 64 + ni * 16 + cl;
        if (col >= N) continue;
        float v = accv[mi][ni][r] + bv[ni];
        if constexpr (PHASE == 0) {
          v = (ACT == 0) ? fmaxf(v, 0.0f)
                         : 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
          reinterpret_cast<ET*>(a.out)[(size_t)e * a.strideOExpert +
                                         (size_t)m * N + col] =
              ETr<ET>::fromf(v);
        } else if constexpr (PHASE == 1) {
          if (multi) {
            // non-atomic per-(token, j) combine slot (summed with the
            // kept mask in k_cast_combine; replaces fp32 atomics)
            reinterpret_cast<ET*>(a.O32)[
                ((size_t)tpsTok(tp.tokenIdx) * a.topk + tpsJ(tp.tokenIdx)) *
                    a.H + col] = ETr<ET>::fromf(v * rowScale);
          } else {
            reinterpret_cast<ET*>(
                a.moe_out)[(size_t)tpsTok(tp.tokenIdx) * a.H + col] =
                ETr<ET>::fromf(v);
          }
        } else if constexpr (PHASE == 2) {  // packed-rows direct output
          reinterpret_cast<ET*>(a.out)[(size_t)e * a.strideOExpert +
                                       (size_t)m * N + col] =
              ETr<ET>::fromf(v);
        } else {  // PHASE 3: gate logits, fp32 (atomic when K-split)
          if (skS > 1)
            atomicAdd(&reinterpret_cast<float*>(a.out)[(size_t)m * N + col], v);
          else
            reinterpret_cast<float*>(a.out)[(size_t)m * N + col] = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Big-tile grouped GEMM, bf16 MFMA: 256xBN (BN 256 or 128), BK=64,
// 512 threads = 8 waves (2M x 4N), per-wave 128 x BN/4 as 8 x BN/64
// mfma_f32_16x16x32_bf16 fragments. Double-buffered glds staging with
// COUNTED vmcnt + raw s_barrier (the guide's deep-pipeline recipe: a
// __syncthreads here would drain vmcnt(0) and expose full HBM latency
// every K-tile - measured 47% wave-park, profiles/r01 PMC). Used for the
// large-M expert shapes; the 128^2 kernel remains for small tiles.
// Same phase semantics/epilogues as k_group_gemm_bf16.
// ---------------------------------------------------------------------------

// effective staging depth for a (BM, BN, WET, STAGES) tile: degrade
// triple buffering to double when it would not fit the 160 KB LDS
template <int BM, int BN, int BEZ, int STAGES>
constexpr int gemm_se() {
  return (STAGES == 3 &&
          3 * BM * 64 * 2 + 3 * BN * 64 * BEZ + BM * 8 + 16 > 160 * 1024)
             ? 2 : STAGES;
}

// LDS arena bytes the tile body needs (mirrored on the host when sizing
// the fused kernel's dynamic arena)
template <int BM, int BN, int BEZ, int STAGES>
constexpr int gemm_lds_bytes() {
  constexpr int SE = gemm_se<BM, BN, BEZ, STAGES>();
  return SE * BM * 64 * 2 + SE * BN * 64 * BEZ + BM * 8 + 16;
}

// ---------------------------------------------------------------------------
// One (e, m0, n0, ksplit) tile job of the big-tile grouped GEMM: the
// shared body of k_group_gemm_bf16_big (classic multi-kernel path) and
// k_moe_fused (single-launch persistent path). act / hasBias are
// runtime here; the classic kernels pass template constants that fold
// after inlining. Layout/state contracts documented at the kernel below.
// ---------------------------------------------------------------------------
template <typename ET, int PHASE, int BN, int BM, typename WET, int STAGES,
          int CK = 0, int CN = 0>
__device__ __forceinline__ bool gemm_job_body(
    const GemmArgs& a, char* smemBase, int e, int ksplit, int m0, int n0,
    int act, bool hasBias) {
  using vec8 = typename ETr<ET>::vec8;
  constexpr int BK = 64;
  constexpr int NF = BN / 64;           // B fragments per wave (4 or 2)
  constexpr int MI = BM / 32;           // A fragments per wave (8 or 4)
  constexpr int BEZ = (int)sizeof(WET);
  constexpr int BCH = BK * BEZ / 16;        // 16B chunks per B row
  constexpr int AGRP = BM * BK * 2 / 1024;  // glds 1KiB groups per A tile
  constexpr int BGRP = BN * BK * BEZ / 1024;
  constexpr int GPW_A = AGRP / 8, GPW_B = BGRP / 8;  // per wave
  constexpr int GPT = GPW_A + GPW_B;    // glds per wave per K-tile
  constexpr int SE = gemm_se<BM, BN, BEZ, STAGES>();
  ET* Abase = reinterpret_cast<ET*>(smemBase);      // SE x [BM][BK]
  WET* Bbase = reinterpret_cast<WET*>(smemBase + SE * BM * BK * 2);
  TPS* sTps = reinterpret_cast<TPS*>(
      smemBase + SE * BM * BK * 2 + SE * BN * BK * BEZ);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // CK/CN: compile-time K/N for the BASELINE shape dictionary (the
  // reference's own architecture freezes these in flashmoe_config.json
  // as compile-time constants); 0 = runtime (generic path). Constant
  // shapes fold the staging address math and fully determine the
  // K-tile trip count (the probe ladder's "template generality" gap).
  const int K = CK ? CK : a.K;
  const int N = CN ? CN : a.N;
  const int sk = a.splitK > 0 ? a.splitK : 1;
  const int kLen = K / sk;            // this split's K range
  const int kStart = ksplit * kLen;

  const TPS* tpsE = a.tokenIds ? a.tokenIds + (size_t)e * a.pEC : nullptr;
  __syncthreads();  // job t-1's epilogue sTps/sScale readers done
  if (tid == 0)
    *sRouted = tpsE ? min(a.eC[e], (uint32_t)a.EC) : (uint32_t)a.nRows;
  __syncthreads();
  const uint32_t routed = *sRouted;
  if ((uint32_t)m0 >= routed) return false;  // empty tile
  const int mCap = a.tokenIds ? a.pEC : a.nRows;  // A-row clamp bound
  const int we = a.segExpert ? a.segExpert[e] : e;  // weight/bias expert
  if (tid < BM) {
    TPS t{0u, 1.0f};
    if ((uint32_t)(m0 + tid) < routed)
      t = tpsE ? tpsE[m0 + tid] : TPS{(uint32_t)(m0 + tid), 1.0f};
    sTps[tid] = t;
  }
  __syncthreads();

  const ET* __restrict__ Ag = reinterpret_cast<const ET*>(a.A);
  const WET* __restrict__ Bg =
      reinterpret_cast<const WET*>(a.B) + (size_t)we * a.strideBExpert;

  // per-lane glds source bases, hoisted out of the K loop
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  // B staging geometry in WET units: 1KiB group = 1024/(BK*BEZ) rows;
  // fp8 swizzle is on the 16B chunk16 index: ^ ((row >> 2) & 3)
  const int bgrow = lane / BCH;
  const int bsc = (lane % BCH) ^ (BEZ == 1 ? ((bgrow >> 2) & 3) : bgrow);
  const int aRowStride = K;  // == a.H (up, logits) / a.K (down) by layout
  const size_t aBase = (size_t)e * a.strideAExpert;  // 0 for the x-gather up phase
  const ET* aSrc[GPW_A];
  const WET* bSrc[GPW_B];
#pragma unroll
  for (int i = 0; i < GPW_A; ++i) {
    const int row = (wave * GPW_A + i) * 8 + grow8;
    const size_t arow = (PHASE == 0) ? (size_t)tpsTok(sTps[row].tokenIdx)
                                     : (size_t)min(m0 + row, mCap - 1);
    aSrc[i] = Ag + aBase + arow * aRowStride + kStart + schunk * 8;
  }
#pragma unroll
  for (int i = 0; i < GPW_B; ++i) {
    const int row = (wave * GPW_B + i) * (1024 / (BK * BEZ)) + bgrow;
    bSrc[i] = Bg + (size_t)min(n0 + row, N - 1) * K + kStart + bsc * (16 / BEZ);
  }

  auto stage = [&](int kt, int buf) {
#pragma unroll
    for (int i = 0; i < GPW_A; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)(aSrc[i] + kt),
          (las_u32*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 512), 16, 0, 0);
#pragma unroll
    for (int i = 0; i < GPW_B; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)(bSrc[i] + kt),
          (las_u32*)((char*)Bbase + buf * BN * BK * BEZ +
                     (wave * GPW_B + i) * 1024), 16, 0, 0);
  };

  const int wr = wave >> 2, wc = wave & 3;  // 2M x 4N wave grid
  f32x4 accv[MI][NF];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) accv[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // one-barrier 2-phase schedule (guide T3 minimum recipe): stage tile
  // t+1 into the other buffer FIRST, compute tile t, then ONE
  // vmcnt(0)+barrier per tile (the drain is cheap: the stage had the
  // whole compute phase to land).
  const int nK = kLen / BK;
  stage(0, 0);
  if constexpr (SE == 3) {
    if (nK > 1) {
      stage(BK, 1);
      wait_vmcnt<GPT>();  // tile 0 landed; tile 1 still in flight
    } else {
      wait_vmcnt<0>();
    }
  } else {
    wait_vmcnt<0>();
  }
  __builtin_amdgcn_s_barrier();
  for (int t = 0; t < nK; ++t) {
    // STAGES==2: staggered staging - waves 0-3 stage tile t+1 before
    // their first MFMA half, waves 4-7 between the halves, so on each
    // SIMD the paired waves run complementary {stage issue | MFMA}
    // segments instead of lockstep (microarch guide, two-waves-per-SIMD).
    // STAGES==3: stage t+2 up front; the latency budget is two compute
    // phases, so no stagger is needed.
    const int pre = t + SE - 1;  // tile staged this iteration
    const bool stageNow = pre < nK;
    // stagger only pays on the wide (NF==4) stream; the BN=128 kernel's
    // shorter MFMA halves lose more to the mid-stream insertion (measured)
    const bool late = (SE == 2) && (NF == 4) && wave >= 4;
    if (stageNow && !late) stage(pre * BK, pre % SE);
    const ET* Al = Abase + (t % SE) * BM * BK;
    const WET* Bl = Bbase + (t % SE) * BN * BK;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      if (s == 1 && late && stageNow) {
        __builtin_amdgcn_s_setprio(0);
        stage(pre * BK, pre % SE);
        __builtin_amdgcn_s_setprio(1);
      }
      vec8 af[MI], bfr[NF];
      const int rl = lane & 15;
      const int cbase = 4 * s + (lane >> 4);
      auto aread = [&](int mi) {
        const int R = wr * (BM / 2) + mi * 16 + rl;
        af[mi] = *reinterpret_cast<const vec8*>(
            &Al[R * BK + ((cbase ^ (R & 7)) * 8)]);
      };
      // Round-robin reads and MFMAs, pinned with sched_barrier(0): each
      // MFMA pair then waits (partial lgkmcnt) only on fragments read a
      // group earlier, instead of the scheduler's whole-segment
      // lgkmcnt(0) drain before the first MFMA (seen in the .s).
#pragma unroll
      for (int ni = 0; ni < NF; ++ni) {
        const int R = wc * (BN / 4) + ni * 16 + rl;
        if constexpr (BEZ == 1) {
          const int c16 = (cbase >> 1) ^ ((R >> 2) & 3);
          const uint32_t* bp = reinterpret_cast<const uint32_t*>(
              (const char*)Bl + (size_t)R * BK + c16 * 16 + (cbase & 1) * 8);
          bfr[ni] = dequant_fp8x8_bf16(bp[0], bp[1]);
        } else {
          bfr[ni] = *reinterpret_cast<const vec8*>(
              &Bl[R * BK + ((cbase ^ (R & 7)) * 8)]);
        }
      }
      aread(0);
      aread(1);
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int g2 = 0; g2 < MI / 2; ++g2) {
        if (g2 < MI / 2 - 1) {
          aread(2 * g2 + 2);
          aread(2 * g2 + 3);
        }
#pragma unroll
        for (int mi = 2 * g2; mi < 2 * g2 + 2; ++mi)
#pragma unroll
          for (int ni = 0; ni < NF; ++ni)
            accv[mi][ni] = ETr<ET>::mfma(af[mi], bfr[ni], accv[mi][ni]);
        __builtin_amdgcn_sched_barrier(0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    if constexpr (SE == 3) {
      // only t+1 must have landed; t+2 (the GPT glds issued this
      // iteration) may still be in flight
      if (stageNow) wait_vmcnt<GPT>(); else wait_vmcnt<0>();
    } else {
      wait_vmcnt<0>();  // t+1 landed
    }
    __builtin_amdgcn_s_barrier();  // readers done AND next tile ready
  }

  // epilogue (same semantics as k_group_gemm_bf16)
  const int cl = lane & 15;
  const int r0 = (lane >> 4) * 4;
  float* sScale = reinterpret_cast<float*>(Abase);
  // slot: write the per-(token, j) combine slot; scaled only at k>1
  // (the reference's k==1 CombineMode::single semantics are unscaled,
  // processor.cuh:173-204)
  const bool slot = (PHASE == 1) && (a.topk > 1 || a.slotAlways);
  const bool scaled = (PHASE == 1) && a.topk > 1;
  if constexpr (PHASE == 1) {
    if (scaled) {
      __syncthreads();
      if (tid < BM) {
        const TPS tp = sTps[tid];
        float sc = 0.0f;
        if ((uint32_t)(m0 + tid) < routed)
          sc = toF(reinterpret_cast<const ET*>(
                   a.gate_out)[(size_t)tpsTok(tp.tokenIdx) * a.PX +
                               a.expertOffset + e]) / tp.probSum;
        sScale[tid] = sc;
      }
      __syncthreads();
    }
  }
  float bv[NF];
#pragma unroll
  for (int ni = 0; ni < NF; ++ni) bv[ni] = 0.f;
  if (hasBias && ksplit == 0) {  // split partials add the bias exactly once
    // per-expert bias slabs when multi-expert (see the 128^2 kernel)
    const ET* bptr = reinterpret_cast<const ET*>(a.bias) +
                     (a.strideBExpert ? (size_t)we * N : 0);
#pragma unroll
    for (int ni = 0; ni < NF; ++ni) {
      const int col = n0 + wc * (BN / 4) + ni * 16 + cl;
      if (col < N) bv[ni] = toF(bptr[col]);
    }
  }
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wr * (BM / 2) + mi * 16 + r0 + r;
      const int m = m0 + row;
      if ((uint32_t)m >= routed) continue;
      const TPS tp = sTps[row];
      const float rowScale = scaled ? sScale[row] : 1.0f;
#pragma unroll
      for (int ni = 0; ni < NF; ++ni) {
        const int col = n0 + wc * (BN / 4) + ni * 16 + cl;
        if (col >= N) continue;
        float v = accv[mi][ni][r] + bv[ni];
        if constexpr (PHASE == 0) {
          v = (act == 0) ? fmaxf(v, 0.0f)
                         : 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
          storeElem(&reinterpret_cast<ET*>(a.out)[(size_t)e * a.strideOExpert +
                                                  (size_t)m * N + col],
                    ETr<ET>::fromf(v), a.sc1Out != 0);
        } else if constexpr (PHASE == 1) {
          if (slot) {
            // non-atomic per-(token, j) combine slot (summed with the
            // kept mask in k_cast_combine; replaces fp32 atomics)
            storeElem(&reinterpret_cast<ET*>(a.O32)[
                          ((size_t)tpsTok(tp.tokenIdx) * a.topk +
                           tpsJ(tp.tokenIdx)) * a.H + col],
                      ETr<ET>::fromf(v * rowScale), a.sc1Out != 0);
          } else {
            reinterpret_cast<ET*>(
                a.moe_out)[(size_t)tpsTok(tp.tokenIdx) * a.H + col] =
                ETr<ET>::fromf(v);
          }
        } else if constexpr (PHASE == 3) {
          // gate logits, fp32; atomic accumulate when the fused gate
          // K-splits the logits GEMM over jobs (logits32 kept zero
          // between forwards by the route's re-zeroing pass)
          if (a.atomicLogits)
            atomicAdd(&reinterpret_cast<float*>(a.out)[(size_t)m * N + col], v);
          else
            reinterpret_cast<float*>(a.out)[(size_t)m * N + col] = v;
        } else {
          reinterpret_cast<ET*>(a.out)[(size_t)e * a.strideOExpert +
                                       (size_t)m * N + col] =
              ETr<ET>::fromf(v);
        }
      }
    }
  }
  return true;
}

// ---------------------------------------------------------------------------
// Big-tile grouped GEMM kernel (classic multi-kernel path): XCD-aware
// remap loop around gemm_job_body. See the body above for the tile
// pipeline; DESIGN.md par.3 for the mode selection.
// ---------------------------------------------------------------------------
template <typename ET, int PHASE, int ACT, bool HAS_BIAS, int BN, int BM = 256,
          typename WET = ET, int STAGES = 2>
__global__ __launch_bounds__(512, 2) void k_group_gemm_bf16_big(GemmArgs a) {
  constexpr int BK = 64;
  constexpr int BEZ = (int)sizeof(WET);
  constexpr int SE = gemm_se<BM, BN, BEZ, STAGES>();
  __shared__ __attribute__((aligned(16))) char smem[
      SE * BM * BK * 2 + SE * BN * BK * BEZ + BM * 8 + 16];

  // XCD-aware block remap (perf only, placement-independent for
  // correctness): the dispatcher places linear block b on XCD b%8; the
  // bijective remap gives each XCD a CONTIGUOUS chunk of (e, nTile)
  // space with mTile fastest, so the blocks sharing a B weight panel
  // (and, for small E, a whole expert) run on one XCD and hit its L2
  // instead of re-reading HBM/L3 (guide T1; staging-traffic-bound at
  // 6.2 TB/s before this, profiles/r01 pmc2).
  const int mT = a.totalJobs > 0 ? a.jobsMT : gridDim.x;
  const int nT = a.totalJobs > 0 ? a.jobsNT : gridDim.y;
  const int gstride = gridDim.x * gridDim.y * gridDim.z;
  const int nBlocks = a.totalJobs > 0 ? a.totalJobs : gstride;
  const int lin0 =
      blockIdx.x + gridDim.x * (blockIdx.y + gridDim.y * blockIdx.z);
  const int qx = nBlocks / 8, rx = nBlocks % 8;
  const int sk = a.splitK > 0 ? a.splitK : 1;
  const int zE = (a.totalJobs > 0 ? a.totalJobs / (mT * nT)
                                  : (int)gridDim.z) / sk;
  // persistent grid (totalJobs > 0): each block walks jobs lin0,
  // lin0+gstride, ... - the per-launch block setup (sTps load, source
  // pointer setup, I$ warm, prologue staging bubble) amortizes over
  // several tiles instead of being paid once per 16-tile block
  // (mfma_probe: the identical tile loop runs 52.5 us/forward-equiv
  // persistent vs 71.4 us launched per-tile at the cfg2 up shape)
  for (int jl = lin0; jl < nBlocks; jl += gstride) {
    const int xcd = jl % 8, pos = jl / 8;
    const int swz = a.noRemap
        ? jl
        : (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
    const int eEff = swz / (mT * nT);  // in [0, E*splitK)
    const int e = eEff % zE;
    const int ksplit = eEff / zE;
    const int rem = swz % (mT * nT);
    (void)gemm_job_body<ET, PHASE, BN, BM, WET, STAGES>(
        a, smem, e, ksplit, (rem % mT) * BM, (rem / mT) * BN, ACT, HAS_BIAS);
  }
}

// ---------------------------------------------------------------------------
// MX-fp8 grouped GEMM (dtype 5): fp8 e4m3 A (runtime-quantized
// activations, per-64-element E8M0 scales) x fp8 e4m3 B (expert weights,
// scale 1) on v_mfma_scale_f32_16x16x128_f8f6f4 - the 2x-rate CDNA4
// path (~4.7 PF/s vs 2.5 bf16). 128x128 tile, BK=128 (one scaled MFMA
// covers the whole K-tile), 512 threads = 8 waves (2M x 4N), fp32
// accumulate, bf16 outputs. Staging geometry is byte-identical to the
// bf16 kernel's (128-B rows, XOR-16B-chunk swizzle, glds double
// buffer); A/B bytes per K-element halve, so HBM/LDS traffic halves
// too. Scale->lane mapping empirically pinned by
// test_mx_mfma_layout_probe: lane-group g's scale byte covers
// k in [64g, 64g+64) (opsel selects the byte; groups 2..3 unused).
// PHASE semantics match gemm_job_body (0 up / 1 down+slots / 2 packed).
// ---------------------------------------------------------------------------

#define MI_OF(BMV) ((BMV) / 32)
template <int PHASE, int BM = 128, int BN = 128, int STAGES = 2>
__device__ __forceinline__ bool mx_gemm_job_body(
    const GemmArgs& a, char* smemBase, int e, int m0, int n0, int act,
    bool hasBias) {
  constexpr int BK = 128;
  // triple buffering (SE=3) gives each stage TWO compute phases of
  // latency budget (the 2-stage pipeline's per-tile vmcnt(0) drain is
  // the measured stall: MFMA pipe ~12%, profiles/r02_mx_pmc.txt);
  // degrade when it would not fit LDS
  constexpr int SE =
      (STAGES == 3 && 3 * (BM + BN) * BK + BM * 8 + 16 > 160 * 1024)
          ? 2 : STAGES;
  uint8_t* Abase = reinterpret_cast<uint8_t*>(smemBase);   // SE x [BM][BK]
  uint8_t* Bbase = reinterpret_cast<uint8_t*>(smemBase + SE * BM * BK);
  TPS* sTps = reinterpret_cast<TPS*>(smemBase + SE * BM * BK + SE * BN * BK);
  uint32_t* sRouted = reinterpret_cast<uint32_t*>(sTps + BM);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int K = a.K, N = a.N;

  const TPS* tpsE = a.tokenIds ? a.tokenIds + (size_t)e * a.pEC : nullptr;
  __syncthreads();  // prior job's LDS readers done
  if (tid == 0)
    *sRouted = tpsE ? min(a.eC[e], (uint32_t)a.EC) : (uint32_t)a.nRows;
  __syncthreads();
  const uint32_t routed = *sRouted;
  if ((uint32_t)m0 >= routed) return false;
  const int mCap = a.tokenIds ? a.pEC : a.nRows;
  const int we = a.segExpert ? a.segExpert[e] : e;
  if (tid < BM) {
    TPS t{0u, 1.0f};
    if ((uint32_t)(m0 + tid) < routed)
      t = tpsE ? tpsE[m0 + tid] : TPS{(uint32_t)(m0 + tid), 1.0f};
    sTps[tid] = t;
  }
  __syncthreads();

  const uint8_t* __restrict__ Ag = reinterpret_cast<const uint8_t*>(a.A);
  const uint8_t* __restrict__ Bg =
      reinterpret_cast<const uint8_t*>(a.B) + (size_t)we * a.strideBExpert;
  const uint8_t* __restrict__ aS =
      reinterpret_cast<const uint8_t*>(a.aScales);

  // glds staging: 16 KiB per operand tile = 16 x 1 KiB groups (8 rows of
  // 128 B each); lane -> (row-in-group = lane/8, 16B chunk = lane%8),
  // chunk XOR-swizzled by row for conflict-free b128 fragment reads
  const int grow8 = lane >> 3;
  const int schunk = (lane & 7) ^ grow8;
  constexpr int GPW_A = BM / 8 / 8;  // 1 KiB glds groups per wave (A)
  constexpr int GPW_B = BN / 8 / 8;
  const size_t aBase = (size_t)e * a.strideAExpert;  // bytes (u8 A)
  const uint8_t* aSrc[GPW_A];
  const uint8_t* bSrc[GPW_B];
#pragma unroll
  for (int i = 0; i < GPW_A; ++i) {
    const int row = (wave * GPW_A + i) * 8 + grow8;
    const size_t arow = (PHASE == 0) ? (size_t)tpsTok(sTps[row].tokenIdx)
                                     : (size_t)min(m0 + row, mCap - 1);
    aSrc[i] = Ag + aBase + arow * (size_t)K + schunk * 16;
  }
#pragma unroll
  for (int i = 0; i < GPW_B; ++i) {
    const int row = (wave * GPW_B + i) * 8 + grow8;
    bSrc[i] = Bg + (size_t)min(n0 + row, N - 1) * K + schunk * 16;
  }
  auto stage = [&](int kt, int buf) {
#pragma unroll
    for (int i = 0; i < GPW_A; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)(aSrc[i] + kt),
          (las_u32*)(Abase + buf * BM * BK + (wave * GPW_A + i) * 1024), 16,
          0, 0);
#pragma unroll
    for (int i = 0; i < GPW_B; ++i)
      __builtin_amdgcn_global_load_lds(
          (gas_u32*)(bSrc[i] + kt),
          (las_u32*)(Bbase + buf * BN * BK + (wave * GPW_B + i) * 1024), 16,
          0, 0);
  };

  // wave grid 2(M) x 4(N): BM/2 rows x BN/4 cols per wave
  const int wr = wave >> 2, wc = wave & 3;
  constexpr int MI = BM / 32, NF = BN / 64;
  f32x4 accv[MI][NF];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) accv[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // per-lane fragment geometry: row rl, k-group g (32 elements each);
  // the lane's SCALE byte must be the contiguous-64-block (g & 1)'s
  // scale (hardware maps scale-group g to the interleaved chunk pair
  // {k>>6 == g&1, (k>>4)&1 == g>>1} - pinned by the MX layout probe)
  const int rl = lane & 15;
  const int gk = lane >> 4;
  const int K64 = K / 64;

  // per-lane A-scale source rows (this lane's 4 fragment rows), hoisted
  const uint8_t* sRow[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    const int R = wr * (BM / 2) + mi * 16 + rl;
    const size_t arow = (PHASE == 0) ? (size_t)tpsTok(sTps[R].tokenIdx)
                                     : (size_t)min(m0 + R, mCap - 1);
    sRow[mi] = aS + (aBase / 64) + arow * K64 + (gk & 1);
  }
  const int nK = K / BK;
  // counted end-of-tile wait: with SE=3, tile t+1 must have landed but
  // tile t+2's GPT glds AND the MI prefetched scale-byte loads (issued
  // after them, completing in order) may stay in flight
  constexpr int GPW_AB = GPW_A + GPW_B;
  constexpr int WN3 = (GPW_AB + MI_OF(BM) <= 10) ? 10 : 12;
  stage(0, 0);
  // software-pipelined scale bytes: tile t's scattered global byte
  // loads are issued one tile AHEAD so they never sit on the MFMA
  // dependency chain
  int sav[MI], savNext[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) sav[mi] = sRow[mi][0];
  if constexpr (SE == 3) {
    if (nK > 1) {
      stage(BK, 1);
      wait_vmcnt<GPW_AB>();  // tile 0 landed; tile 1 in flight
    } else {
      wait_vmcnt<0>();
    }
  } else {
    wait_vmcnt<0>();
  }
  __builtin_amdgcn_s_barrier();
  for (int t = 0; t < nK; ++t) {
    const int pre = t + SE - 1;      // tile staged this iteration
    const bool stageNow = pre < nK;
    if (stageNow) stage(pre * BK, pre % SE);
    if (t + 1 < nK) {
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) savNext[mi] = sRow[mi][(t + 1) * 2];
    }
    const uint8_t* Al = Abase + (t % SE) * BM * BK;
    const uint8_t* Bl = Bbase + (t % SE) * BN * BK;
    __builtin_amdgcn_s_setprio(1);
    i32x8 af[MI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
      const int R = wr * (BM / 2) + mi * 16 + rl;
      const int c0 = (2 * gk) ^ (R & 7);
      const int c1 = (2 * gk + 1) ^ (R & 7);
      const u32x4 lo = *reinterpret_cast<const u32x4*>(&Al[R * BK + c0 * 16]);
      const u32x4 hi = *reinterpret_cast<const u32x4*>(&Al[R * BK + c1 * 16]);
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        af[mi][d] = (int)lo[d];
        af[mi][4 + d] = (int)hi[d];
      }
    }
    i32x8 bfr[NF];
#pragma unroll
    for (int ni = 0; ni < NF; ++ni) {
      const int R = wc * (BN / 4) + ni * 16 + rl;
      const int c0 = (2 * gk) ^ (R & 7);
      const int c1 = (2 * gk + 1) ^ (R & 7);
      const u32x4 lo = *reinterpret_cast<const u32x4*>(&Bl[R * BK + c0 * 16]);
      const u32x4 hi = *reinterpret_cast<const u32x4*>(&Bl[R * BK + c1 * 16]);
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        bfr[ni][d] = (int)lo[d];
        bfr[ni][4 + d] = (int)hi[d];
      }
    }
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NF; ++ni)
        accv[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            af[mi], bfr[ni], accv[mi][ni], 0 /*fp8*/, 0 /*fp8*/, 0, sav[mi],
            0, 0x7F /*B scale 1.0*/);
    __builtin_amdgcn_s_setprio(0);
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) sav[mi] = savNext[mi];
    if constexpr (SE == 3) {
      if (stageNow) wait_vmcnt<WN3>(); else wait_vmcnt<0>();
    } else {
      wait_vmcnt<0>();
    }
    __builtin_amdgcn_s_barrier();
  }

  // epilogue (bf16 outputs; mirrors gemm_job_body's PHASE semantics)
  const int cl = lane & 15;
  const int r0 = (lane >> 4) * 4;
  float* sScale = reinterpret_cast<float*>(Abase);
  const bool slot = (PHASE == 1) && (a.topk > 1 || a.slotAlways);
  const bool scaled = (PHASE == 1) && a.topk > 1;
  if constexpr (PHASE == 1) {
    if (scaled) {
      __syncthreads();
      if (tid < BM) {
        const TPS tp = sTps[tid];
        float sc = 0.0f;
        if ((uint32_t)(m0 + tid) < routed)
          sc = toF(reinterpret_cast<const bf16*>(
                   a.gate_out)[(size_t)tpsTok(tp.tokenIdx) * a.PX +
                               a.expertOffset + e]) / tp.probSum;
        sScale[tid] = sc;
      }
      __syncthreads();
    }
  }
  float bv[NF];
#pragma unroll
  for (int ni = 0; ni < NF; ++ni) bv[ni] = 0.f;
  if (hasBias) {
    const bf16* bptr = reinterpret_cast<const bf16*>(a.bias) +
                       (a.strideBExpert ? (size_t)we * N : 0);
#pragma unroll
    for (int ni = 0; ni < NF; ++ni) {
      const int col = n0 + wc * (BN / 4) + ni * 16 + cl;
      if (col < N) bv[ni] = toF(bptr[col]);
    }
  }
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = wr * (BM / 2) + mi * 16 + r0 + r;
      const int m = m0 + row;
      if ((uint32_t)m >= routed) continue;
      const TPS tp = sTps[row];
      const float rowScale = scaled ? sScale[row] : 1.0f;
#pragma unroll
      for (int ni = 0; ni < NF; ++ni) {
        const int col = n0 + wc * (BN / 4) + ni * 16 + cl;
        if (col >= N) continue;
        float v = accv[mi][ni][r] + bv[ni];
        if constexpr (PHASE == 0) {
          v = (act == 0) ? fmaxf(v, 0.0f)
                         : 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
          if (!(a.out8 && BN == 256))
            reinterpret_cast<bf16*>(a.out)[(size_t)e * a.strideOExpert +
                                           (size_t)m * N + col] =
                __float2bfloat16(v);
        } else if constexpr (PHASE == 1) {
          if (slot) {
            reinterpret_cast<bf16*>(a.O32)[
                ((size_t)tpsTok(tp.tokenIdx) * a.topk + tpsJ(tp.tokenIdx)) *
                    a.H + col] = __float2bfloat16(v * rowScale);
          } else {
            reinterpret_cast<bf16*>(
                a.moe_out)[(size_t)tpsTok(tp.tokenIdx) * a.H + col] =
                __float2bfloat16(v);
          }
        } else {
          reinterpret_cast<bf16*>(a.out)[(size_t)e * a.strideOExpert +
                                         (size_t)m * N + col] =
              __float2bfloat16(v);
        }
      }
    }
  }
  if constexpr (PHASE == 0 && BN == 256) {
    if (a.out8) {
      // epilogue MX quantization: each quarter-wave (16 lanes x NF=4
      // cols each) holds exactly ONE contiguous 64-col block of a row.
      // bf16-round first (the CPU restatement Element-rounds the
      // intermediate before quantizing), 16-lane blockmax, E8M0 scale,
      // e4m3 RNE pack via v_cvt_pk_fp8_f32.
      uint8_t* o8 = reinterpret_cast<uint8_t*>(a.out8) +
                    (size_t)e * a.strideOExpert;
      uint8_t* oS = reinterpret_cast<uint8_t*>(a.outScales) +
                    (size_t)e * (a.strideOExpert / 64);
      const int blk = n0 / 64 + wc;  // this quarter-wave's 64-block
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr * (BM / 2) + mi * 16 + r0 + r;
          const int m = m0 + row;
          if ((uint32_t)m >= routed) continue;
          float vq[NF];
          float mx = 0.0f;
#pragma unroll
          for (int ni = 0; ni < NF; ++ni) {
            float v = accv[mi][ni][r] + bv[ni];
            v = (act == 0)
                    ? fmaxf(v, 0.0f)
                    : 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
            vq[ni] = __bfloat162float(__float2bfloat16(v));
            mx = fmaxf(mx, fabsf(vq[ni]));
          }
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            mx = fmaxf(mx, __shfl_xor(mx, off, 16));
          int eE = 0;
          if (mx > 0.0f) {
            int ex;
            (void)frexpf(mx / 448.0f, &ex);
            eE = ex < -126 ? -126 : (ex > 127 ? 127 : ex);
          }
          const float inv = exp2f((float)-eE);
          const uint32_t p01 = cvt2_fp8(vq[0] * inv, vq[1] * inv);
          const uint32_t p23 = cvt2_fp8(vq[2] * inv, vq[3] * inv);
          uint8_t* orow = o8 + (size_t)m * N;
#pragma unroll
          for (int ni = 0; ni < NF; ++ni) {
            const int col = n0 + wc * (BN / 4) + ni * 16 + cl;
            const uint32_t src = (ni < 2) ? p01 : p23;
            orow[col] = (uint8_t)((src >> ((ni & 1) * 8)) & 0xFF);
          }
          if (cl == 0)
            oS[(size_t)m * (N / 64) + blk] = (uint8_t)(127 + eE);
        }
      }
    }
  }
  return true;
}

template <int PHASE, int ACT, bool HAS_BIAS, int BM = 128, int BN = 128,
          int STAGES = 2>
__global__ __launch_bounds__(512, 2) void k_group_gemm_mx(GemmArgs a) {
  constexpr int BK = 128;
  constexpr int SE =
      (STAGES == 3 && 3 * (BM + BN) * BK + BM * 8 + 16 > 160 * 1024)
          ? 2 : STAGES;
  __shared__ __attribute__((aligned(16))) char smem[
      SE * BM * BK + SE * BN * BK + BM * 8 + 16];
  const int mT = a.totalJobs > 0 ? a.jobsMT : gridDim.x;
  const int nT = a.totalJobs > 0 ? a.jobsNT : gridDim.y;
  const int gstride = gridDim.x * gridDim.y * gridDim.z;
  const int nBlocks = a.totalJobs > 0 ? a.totalJobs : gstride;
  const int lin0 =
      blockIdx.x + gridDim.x * (blockIdx.y + gridDim.y * blockIdx.z);
  const int qx = nBlocks / 8, rx = nBlocks % 8;
  for (int jl = lin0; jl < nBlocks; jl += gstride) {
    const int xcd = jl % 8, pos = jl / 8;
    const int swz = a.noRemap
        ? jl
        : (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
    const int e = swz / (mT * nT);
    const int rem = swz % (mT * nT);
    (void)mx_gemm_job_body<PHASE, BM, BN, STAGES>(
        a, smem, e, (rem % mT) * BM, (rem / mT) * BN, ACT, HAS_BIAS);
  }
}

// ---------------------------------------------------------------------------
// fp32 grouped GEMM (config-1 "CPU correctness plumbing" shapes; VALU
// tiled). 64x64 tile, 256 threads, each thread a 4x4 sub-block, BK=16.
// Same phase semantics as the bf16 kernel.
// ---------------------------------------------------------------------------

template <int PHASE, int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(256) void k_group_gemm_f32(GemmArgs a) {
  constexpr int BM = 64, BN = 64, BK = 16;
  constexpr int LDT = BK + 1;
  __shared__ float Alds[BM * LDT];
  __shared__ float Blds[BN * LDT];
  __shared__ TPS sTps[BM];
  __shared__ uint32_t sRouted;

  const int e = blockIdx.z;
  const int tid = threadIdx.x;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int K = a.K, N = a.N;
  const TPS* tpsE = a.tokenIds ? a.tokenIds + (size_t)e * a.pEC : nullptr;
  const int we = a.segExpert ? a.segExpert[e] : e;
  if (tid == 0)
    sRouted = tpsE ? min(a.eC[e], (uint32_t)a.EC) : (uint32_t)a.nRows;
  __syncthreads();
  const uint32_t routed = sRouted;
  if ((uint32_t)m0 >= routed) return;
  if (tid < BM) {
    TPS t{0u, 1.0f};
    if ((uint32_t)(m0 + tid) < routed)
      t = tpsE ? tpsE[m0 + tid] : TPS{(uint32_t)(m0 + tid), 1.0f};
    sTps[tid] = t;
  }
  __syncthreads();

  const float* __restrict__ Ag = reinterpret_cast<const float*>(a.A);
  const float* __restrict__ Bg =
      reinterpret_cast<const float*>(a.B) + (size_t)we * a.strideBExpert;

  const int tr = (tid / 16) * 4;  // thread rows [tr, tr+4)
  const int tc = (tid % 16) * 4;
  float acc[4][4] = {};
  for (int kt = 0; kt < K; kt += BK) {
    // stage: BM*BK = 1024 floats, 256 threads x 4
#pragma unroll
    for (int u4 = 0; u4 < 4; ++u4) {
      const int u = tid + u4 * 256;
      const int row = u / BK, cc = u % BK;
      const float* src = (PHASE == 0)
          ? Ag + (size_t)tpsTok(sTps[row].tokenIdx) * a.H + kt + cc
          : Ag + (size_t)e * a.strideAExpert + (size_t)(m0 + row) * K + kt + cc;
      Alds[row * LDT + cc] = *src;
      const int brow = min(n0 + row, N - 1);
      Blds[row * LDT + cc] = Bg[(size_t)brow * K + kt + cc];
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; ++kk) {
      float av[4], bv[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) av[i] = Alds[(tr + i) * LDT + kk];
#pragma unroll
      for (int j = 0; j < 4; ++j) bv[j] = Blds[(tc + j) * LDT + kk];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = fmaf(av[i], bv[j], acc[i][j]);
    }
    __syncthreads();
  }

  const bool multi = (PHASE == 1) && a.topk > 1;
  float bv[4] = {0.f, 0.f, 0.f, 0.f};
  if constexpr (HAS_BIAS) {
    // per-expert bias slabs when multi-expert (see the 128^2 kernel)
    const float* bptr = reinterpret_cast<const float*>(a.bias) +
                        (a.strideBExpert ? (size_t)we * N : 0);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = n0 + tc + j;
      if (col < N) bv[j] = bptr[col];
    }
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int m = m0 + tr + i;
    if ((uint32_t)m >= routed) continue;
    const TPS tp = sTps[tr + i];
    float rowScale = 1.0f;
    if (multi)
      rowScale = reinterpret_cast<const float*>(
                     a.gate_out)[(size_t)tpsTok(tp.tokenIdx) * a.PX +
                                 a.expertOffset + e] / tp.probSum;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = n0 + tc + j;
      if (col >= N) continue;
      float v = acc[i][j] + bv[j];
      if constexpr (PHASE == 0) {
        v = (ACT == 0) ? fmaxf(v, 0.0f)
                       : 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
        reinterpret_cast<float*>(a.out)[(size_t)e * a.strideOExpert +
                                        (size_t)m * N + col] = v;
      } else if constexpr (PHASE == 1) {
        if (multi) {
          reinterpret_cast<float*>(a.O32)[
              ((size_t)tpsTok(tp.tokenIdx) * a.topk + tpsJ(tp.tokenIdx)) *
                  a.H + col] = v * rowScale;
        } else {
          reinterpret_cast<float*>(
              a.moe_out)[(size_t)tpsTok(tp.tokenIdx) * a.H + col] = v;
        }
      } else {
        reinterpret_cast<float*>(a.out)[(size_t)e * a.strideOExpert +
                                        (size_t)m * N + col] = v;
      }
    }
  }
}

// sum the k non-atomic combine slots (kept-masked) into the output
// (k>1 path; replaces the fp32-atomic accumulator + cast). One block
// per token chunk, 16-B vector loads per slot (a scalar grid-stride
// version with a per-element div ran at 1.35 TB/s on the cfg5 shape).
template <typename T, int K>
__global__ void k_cast_combine(const T* __restrict__ cbuf,
                               const uint8_t* __restrict__ kept,
                               T* __restrict__ out, int S, int H) {
  constexpr int EPU = 16 / sizeof(T);
  for (int t = blockIdx.x; t < S; t += gridDim.x) {
    bool kp[K];
#pragma unroll
    for (int j = 0; j < K; ++j) kp[j] = kept[(size_t)t * K + j] != 0;
    for (int h = threadIdx.x * EPU; h < H; h += blockDim.x * EPU) {
      float acc[EPU];
#pragma unroll
      for (int q = 0; q < EPU; ++q) acc[q] = 0.0f;
#pragma unroll
      for (int j = 0; j < K; ++j) {
        if (!kp[j]) continue;
        const u32x4 v = *reinterpret_cast<const u32x4*>(
            cbuf + ((size_t)t * K + j) * H + h);
#pragma unroll
        for (int w = 0; w < 4; ++w) {
          const uint32_t vw = v[w];
          if constexpr (__is_same(T, bf16)) {
            const float2 f = __bfloat1622float2(
                *reinterpret_cast<const __hip_bfloat162*>(&vw));
            acc[2 * w] += f.x;
            acc[2 * w + 1] += f.y;
          } else if constexpr (__is_same(T, fp16)) {
            const float2 f = __half22float2(
                *reinterpret_cast<const __half2*>(&vw));
            acc[2 * w] += f.x;
            acc[2 * w + 1] += f.y;
          } else {
            acc[w] += __uint_as_float(vw);
          }
        }
      }
      T ov[EPU];
#pragma unroll
      for (int q = 0; q < EPU; ++q) fromF(acc[q], ov[q]);
      *reinterpret_cast<u32x4*>(out + (size_t)t * H + h) =
          *reinterpret_cast<const u32x4*>(ov);
    }
  }
}

// cast the fp32 combine accumulator into the Element output (k>1 path)
// and RE-ZERO it for the next forward (O32 is zeroed once at initialize;
// keeping it clean here saves a 16 MB memset pass per call)
template <typename T>
__global__ void k_cast_out(float* __restrict__ O32, T* __restrict__ out,
                           size_t n) {
  const size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t j = i; j < n; j += stride) {
    fromF(O32[j], out[j]);
    O32[j] = 0.0f;
  }
}

// combine pre-packed return rows (EP host path): one block per row
template <typename T, bool SCALED>
__global__ void k_combine_rows(const T* __restrict__ rows,
                               const uint32_t* __restrict__ tokenIdx,
                               const float* __restrict__ scale,
                               float* __restrict__ O32, T* __restrict__ moeOut,
                               int H) {
  const int i = blockIdx.x;
  const uint32_t t = tokenIdx[i];
  const float s = SCALED ? scale[i] : 1.0f;
  for (int h = threadIdx.x; h < H; h += blockDim.x) {
    const float v = toF(rows[(size_t)i * H + h]);
    if (SCALED)
      atomicAdd(&O32[(size_t)t * H + h], s * v);
    else
      moeOut[(size_t)t * H + h] = rows[(size_t)i * H + h];
  }
}

// pack the dispatch send buffer: send[e][i][:] = x[tokenIds[e][i]][:]
// for i < min(eC[e], EC); rows past the routed count keep whatever is
// there (they are dropped at the source on combine). This is the
// reference's symmetric-heap cell layout (peer-major expert-slot cells,
// types.cuh:1014-1032) built for a capacity-padded static all_to_all.
template <typename T>
__global__ void k_pack_dispatch(const T* __restrict__ x,
                                const TPS* __restrict__ tokenIds,
                                const uint32_t* __restrict__ eC,
                                T* __restrict__ sendbuf, int H, int EC,
                                int pEC) {
  const int e = blockIdx.y;
  const int i = blockIdx.x;
  const uint32_t r = min(eC[e], (uint32_t)EC);
  if ((uint32_t)i >= r) return;
  const uint32_t tok = tpsTok(tokenIds[(size_t)e * pEC + i].tokenIdx);
  constexpr int EPU = 16 / sizeof(T);  // elements per 16B unit
  const T* src = x + (size_t)tok * H;
  T* dst = sendbuf + ((size_t)e * EC + i) * H;
  for (int h = threadIdx.x * EPU; h < H; h += blockDim.x * EPU) {
    *reinterpret_cast<u32x4*>(dst + h) =
        *reinterpret_cast<const u32x4*>(src + h);
  }
}

// combine the capacity-padded RETURNED buffer at the source: for each
// (e, i < routed), out[token] += gate_out[token, e]/probSum * row
// (k>1; k==1 unscaled overwrite semantics via scale=1 on a zero
// accumulator). All metadata is source-local (tokenIds/eC/gate_out) -
// no host involvement (processor.cuh:44-205 combine semantics).
template <typename T>
__global__ void k_combine_padded(const T* __restrict__ rows,
                                 const TPS* __restrict__ tokenIds,
                                 const uint32_t* __restrict__ eC,
                                 const T* __restrict__ gate_out,
                                 T* __restrict__ cbuf, int H, int EC,
                                 int pEC, int PX, int topk) {
  const int e = blockIdx.y;
  const int i = blockIdx.x;
  const uint32_t r = min(eC[e], (uint32_t)EC);
  if ((uint32_t)i >= r) return;
  const TPS tp = tokenIds[(size_t)e * pEC + i];
  const uint32_t tok = tpsTok(tp.tokenIdx);
  const uint32_t j = tpsJ(tp.tokenIdx);
  const float sc = (topk > 1)
      ? toF(gate_out[(size_t)tok * PX + e]) / tp.probSum
      : 1.0f;
  const T* src = rows + ((size_t)e * EC + i) * H;
  // non-atomic per-(token, j) combine slot (k_cast_combine sums them)
  T* dst = cbuf + ((size_t)tok * topk + j) * H;
  for (int h = threadIdx.x; h < H; h += blockDim.x) {
    fromF(sc * toF(src[h]), dst[h]);
  }
}

// ---------------------------------------------------------------------------
// One-sided P2P dispatch/return over xGMI (the reference's intra-node
// mode: direct stores into peers' symmetric-heap cells + system-scope
// 8-byte signals, os/packet.cuh:214-258 / bootstrap.cuh:442-443
// nvshmem_ptr resolution). Heap cell layout identical to the padded
// all_to_all path ([source, local expert, slot<EC, H]); flags are
// seq-tagged so they never need re-zeroing (seqBit concept,
// types.cuh:1045-1063). Visibility protocol per the CDNA4 guide
// (Guideline 16, system scope for cross-device): payload stores ->
// per-wave vmcnt drain -> one-lane system-scope release fence + asm
// vmcnt drain -> relaxed system-scope flag store; consumer polls
// relaxed, then one system-scope acquire fence.
// ---------------------------------------------------------------------------

// write this rank's routed rows for expert e straight into the OWNER
// rank's recv heap cell [myRank][e%nLx][i]; the last arriving block of
// each expert signals the owner's dispatch flag with `seq`.
template <typename T>
__global__ void k_dispatch_p2p(const T* __restrict__ x,
                               const TPS* __restrict__ tokenIds,
                               const uint32_t* __restrict__ eC,
                               const uint64_t* __restrict__ peerRecv,
                               const uint64_t* __restrict__ peerDispFlags,
                               uint32_t* __restrict__ arrive, int H, int EC,
                               int pEC, int nLx, int myRank,
                               unsigned long long seq) {
  const int e = blockIdx.y;
  const int i = blockIdx.x;
  const uint32_t r = min(eC[e], (uint32_t)EC);
  const uint32_t participants = max(r, 1u);
  if ((uint32_t)i >= participants) return;
  const int owner = e / nLx;
  const int le = e % nLx;
  T* heap = reinterpret_cast<T*>(peerRecv[owner]);
  if ((uint32_t)i < r) {
    const uint32_t tok = tpsTok(tokenIds[(size_t)e * pEC + i].tokenIdx);
    const T* src = x + (size_t)tok * H;
    T* dst = heap + (((size_t)myRank * nLx + le) * EC + i) * H;
    constexpr int EPU = 16 / sizeof(T);
    for (int h = threadIdx.x * EPU; h < H; h += blockDim.x * EPU) {
      *reinterpret_cast<u32x4*>(dst + h) =
          *reinterpret_cast<const u32x4*>(src + h);
    }
  }
  // every wave drains its own stores before the block arrives
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    const uint32_t got = __hip_atomic_fetch_add(
        arrive + e, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (got + 1 == participants) {
      // last arriver publishes to the owner: system release + flag
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      unsigned long long* flag =
          reinterpret_cast<unsigned long long*>(peerDispFlags[owner]) +
          (size_t)myRank * nLx + le;
      __hip_atomic_store(flag, seq, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

// wait until every (source, local expert) dispatch flag equals seq.
// On timeout the device error word (host-mapped, zero-copy) is set so
// the host can detect the poisoned exchange without a stream sync:
// fm_dispatch_p2p/fm_return_p2p check it at entry and
// fm_p2p_error_check() checks it definitively after a sync.
__global__ void k_await_flags(const unsigned long long* __restrict__ flags,
                              int n, unsigned long long seq,
                              uint32_t* __restrict__ err, long long maxSpin) {
  bool done = false;
  for (long long spin = 0; !done && spin < maxSpin; ++spin) {
    done = true;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
      done &= (__hip_atomic_load(flags + i, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_SYSTEM) >= seq);
    }
    done = __syncthreads_and(done);
    if (!done) __builtin_amdgcn_s_sleep(16);
  }
  if (threadIdx.x == 0 && !done) {
    // bounded spin gave up: poison is preferable to a hang, but the
    // host MUST be able to see it happened
    printf("flashmoe: k_await_flags timeout (seq %llu)\n", seq);
    if (err)
      __hip_atomic_store(err, (uint32_t)seq ? (uint32_t)seq : 1u,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
  __syncthreads();
}

// return the FFN results: segment (s, le) rows go back to source rank
// s's return heap at cell [global expert][i]; last block per segment
// signals the source's return flag for that global expert.
template <typename T>
__global__ void k_return_p2p(const T* __restrict__ ffn_out,
                             const uint64_t* __restrict__ peerRet,
                             const uint64_t* __restrict__ peerRetFlags,
                             uint32_t* __restrict__ arrive, int H, int EC,
                             int nLx, int myRank,
                             unsigned long long seq) {
  const int seg = blockIdx.y;  // (source rank, local expert)
  const int i = blockIdx.x;
  const int src = seg / nLx;
  const int le = seg % nLx;
  const int ge = myRank * nLx + le;  // global expert id (I am the owner)
  T* ret = reinterpret_cast<T*>(peerRet[src]);
  const T* row = ffn_out + ((size_t)seg * EC + i) * H;
  T* dst = ret + ((size_t)ge * EC + i) * H;
  constexpr int EPU = 16 / sizeof(T);
  for (int h = threadIdx.x * EPU; h < H; h += blockDim.x * EPU) {
    *reinterpret_cast<u32x4*>(dst + h) =
        *reinterpret_cast<const u32x4*>(row + h);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    const uint32_t got = __hip_atomic_fetch_add(
        arrive + seg, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (got + 1 == (uint32_t)EC) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      unsigned long long* flag =
          reinterpret_cast<unsigned long long*>(peerRetFlags[src]) + ge;
      __hip_atomic_store(flag, seq, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

// export clipped routing into caller tensors: routed[e] = min(eC, EC),
// out[e][i] = {tokenIdx, probSum-bits} for i < routed (EP host pipeline)
__global__ void k_export_routing(const TPS* __restrict__ tokenIds,
                                 const uint32_t* __restrict__ eC,
                                 uint32_t* __restrict__ routed,
                                 uint32_t* __restrict__ out, int E, int EC,
                                 int pEC) {
  const int e = blockIdx.x;
  const uint32_t r = min(eC[e], (uint32_t)EC);
  if (threadIdx.x == 0) routed[e] = r;
  for (int i = threadIdx.x; i < EC; i += blockDim.x) {
    TPS t = (i < (int)r) ? tokenIds[(size_t)e * pEC + i] : TPS{0u, 0.0f};
    out[((size_t)e * EC + i) * 2] = tpsTok(t.tokenIdx);
    out[((size_t)e * EC + i) * 2 + 1] = __float_as_uint(t.probSum);
  }
}

// MX-scaled MFMA layout probe (test-only): D = block-scaled A[16x128] x
// B[128x16] via one mfma_scale_f32_16x16x128_f8f6f4 (fp8 e4m3 A and B,
// E8M0 scales). Empirically pinned contract (test_mx_mfma_layout_probe):
// lane l holds A[row=l&15][k=32*(l>>4)..+32] and B[...][col=l&15] as 8
// dwords; the scale byte of lane-group g covers the interleaved chunk
// pair {k: k>>6 == g&1, (k>>4)&1 == g>>1} (so per-contiguous-64 scales
// are supplied as block (g&1)); value 2^(byte-127); C/D as every 16x16.
template <int OPS>
__global__ void k_mx_mfma_probe(const uint8_t* A, const uint8_t* B,
                                const uint8_t* sa, const uint8_t* sb,
                                float* D) {
  const int lane = threadIdx.x & 63;
  const int row = lane & 15;
  const int kb = lane >> 4;  // 32-element K block (assumed data layout)
  i32x8 af, bf;
#pragma unroll
  for (int d = 0; d < 8; ++d) {
    af[d] = *reinterpret_cast<const int*>(&A[row * 128 + kb * 32 + d * 4]);
    bf[d] = *reinterpret_cast<const int*>(&B[row * 128 + kb * 32 + d * 4]);
    // B given transposed: B_t[col][k] with col=lane&15 - same indexing
  }
  f32x4 c{0.f, 0.f, 0.f, 0.f};
  // sa/sb are PER-LANE raw scale i32s (64 each): the GPU layout test
  // sweeps single lanes/bytes to pin the hardware's scale->block map
  const int sav = reinterpret_cast<const int*>(sa)[lane];
  const int sbv = reinterpret_cast<const int*>(sb)[lane];
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(af, bf, c, 0, 0,
                                                       OPS, sav, OPS, sbv);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

// MFMA layout probe (test-only): D = A[16x32] x B[32x16] via one
// mfma_f32_16x16x32_bf16, written with the assumed C/D mapping.
__global__ void k_fp8cvt_probe(const uint32_t* in, float* out) {
  const int t = threadIdx.x;  // 32 threads x 8 values
  const bf16x8 v = dequant_fp8x8_bf16(in[t * 2], in[t * 2 + 1]);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[t * 8 + i] = (float)v[i];
}

__global__ void k_mfma_probe(const bf16* A, const bf16* B, float* D) {
  const int lane = threadIdx.x & 63;
  bf16x8 af, bfr;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = *reinterpret_cast<const __bf16*>(
        &A[(lane & 15) * 32 + (lane >> 4) * 8 + j]);
    bfr[j] = *reinterpret_cast<const __bf16*>(
        &B[(lane & 15) * 32 + (lane >> 4) * 8 + j]);  // B given as [16 col][32 k]
  }
  f32x4 c{0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

// ===========================================================================
// Fused persistent forward kernel (the north star's single-launch DMoE,
// reference moe::forward, moe.cuh:77-144): ONE launch whose resident
// blocks walk four job phases -
//   L: gate-logits GEMM tiles (K-split, fp32 atomic accumulate into
//      logits32) with per-tile arrival counters; the LAST arriving block
//      of a tile runs the softmax/top-k/route for that tile inline
//      (gate.cuh:474-720 semantics, fused per VERDICT r01 #4)
//   U: expert up-GEMM tiles (act(x W_up^T + b), token-gather A)
//   D: expert down-GEMM tiles with the combine-slot epilogue
//   C: masked slot reduction into moe_out
// Phase seams are job-done fan-ins + one release per producing block,
// following the CDNA4 guide's Guideline 16 protocol exactly: payload
// plain stores -> every wave s_waitcnt vmcnt(0) -> __syncthreads ->
// lane 0 agent-scope release fence + asm vmcnt drain -> relaxed counter
// add; consumers poll relaxed, then ONE agent acquire + __syncthreads.
// All polled words are zeroed by hipMemsetAsync nodes ahead of the
// launch (replayed first under hipGraph capture). Every spin is bounded
// and a give-up sets the host-mapped error word instead of hanging.
// The grid is sized to exactly the co-resident block count (1 block/CU
// at the 146 KB LDS arena), so the in-kernel waits cannot deadlock.
// ===========================================================================

struct FusedCtl {
  uint32_t routeTilesDone;   // fan-in: tiles fully routed
  uint32_t upJobsDone;       // fan-in: up-GEMM jobs completed
  uint32_t dnJobsDone;       // fan-in: down-GEMM jobs completed
  uint32_t pad0;
  unsigned long long clk[6]; // atomicMax phase-end wall clocks:
                             // 0 entry, 1 route done, 2 up, 3 down, 4 end
  // followed in the same allocation by tileArrive[nTiles]
};

__device__ __forceinline__ void fused_release_arrive(uint32_t* ctr) {
  // producer side of the Guideline 16 hand-off (counter form)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    // restate the post-wbl2 drain where the compiler would drop it
    // (guide Guideline 16 pitfall 12)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __hip_atomic_fetch_add(ctr, 1u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
  }
}

__device__ __forceinline__ void fused_wait(const uint32_t* ctr,
                                           uint32_t target,
                                           uint32_t* errWord,
                                           long long bound,
                                           bool noAcq = false) {
  if (threadIdx.x == 0) {
    long long spins = 0;
    while (__hip_atomic_load(ctr, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT) < target) {
      __builtin_amdgcn_s_sleep(64);
      if (++spins > bound) {
        // give up: poisoned output, host sees the error word
        if (errWord)
          __hip_atomic_store(errWord, 2u, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM);
        break;
      }
    }
    if (!noAcq) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
}

struct FusedMeta {
  FusedCtl* ctl;
  uint32_t* tileArrive;      // [nTiles] logits-job arrivals per tile
  uint32_t* upRowDone;       // [E, upMT] completed up n-tiles per row-tile
  int upBM, upRowTarget;     // up tile rows / n-tiles per complete row
  uint32_t* eCf;             // per-expert routed counts (g.eC)
  float* logits32;
  void* gate_out;
  TPS* tokenIds;
  const void* cbuf;          // combine slots [S, k, H] Element
  uint8_t* kept;
  void* moe_out;
  float* gML;                // training aux (null in inference)
  float* gMeC;
  uint32_t* errWord;         // host-mapped (system scope)
  int S, H, E, PX, EC, pEC, topk, training;
  int nTiles;                // S / 128
  int lgKS, lgNT, lgJobs;    // logits K-split, N tiles, total jobs
  int upGeom, upMT, upNT, upJobs;
  int dnGeom, dnMT, dnNT, dnJobs;
  int cbTB, cbJobs;          // combine token-block size / jobs
  int debugBits;             // FM_FUSED_DEBUG bit0: skip per-job drains
                             // (publish once at walk end), bit1: skip
                             // acquire fences. MEASUREMENT ONLY - both
                             // break the visibility protocol; never set
                             // outside an A/B pricing run.
  int nBlocks;
  int ctlOff;                // LDS offset of the 16-B control slice
  long long spinBound;
};

// softmax / top-k / ordered placement for one 128-token tile inside the
// fused kernel: thin wrapper over route_tile_core (shared with the
// classic k_gate_route)
template <typename T, int K>
__device__ __forceinline__ void route_tile(char* smem, const FusedMeta& f,
                                           int tile) {
  route_tile_core<T, K>(smem, f.logits32, reinterpret_cast<T*>(f.gate_out),
                        f.tokenIds, f.eCf, f.kept, f.gML, f.gMeC, f.S, f.E,
                        f.PX, f.EC, f.pEC, tile * 128);
}

// masked slot reduction (k_cast_combine semantics) over a token block
template <typename T, int K>
__device__ __forceinline__ void combine_tokens(const FusedMeta& f, int t0,
                                               int t1) {
  constexpr int EPU = 16 / sizeof(T);
  const T* cbuf = reinterpret_cast<const T*>(f.cbuf);
  T* out = reinterpret_cast<T*>(f.moe_out);
  const int H = f.H;
  // flat (token, h-unit) sweep: keeps all 512 threads busy even when
  // H/EPU < 512 (the per-token h loop left 3/4 of the block idle)
  const int units = H / EPU;
  const int total = (t1 - t0) * units;
  for (int u = threadIdx.x; u < total; u += 512) {
    const int t = t0 + u / units;
    const int h = (u % units) * EPU;
    {
      float acc[EPU];
#pragma unroll
      for (int q = 0; q < EPU; ++q) acc[q] = 0.0f;
#pragma unroll
      for (int j = 0; j < K; ++j) {
        if (f.kept[(size_t)t * K + j] == 0) continue;
        const u32x4 v = *reinterpret_cast<const u32x4*>(
            cbuf + ((size_t)t * K + j) * H + h);
#pragma unroll
        for (int w = 0; w < 4; ++w) {
          const uint32_t vw = v[w];
          if constexpr (__is_same(T, bf16)) {
            const float2 fv = __bfloat1622float2(
                *reinterpret_cast<const __hip_bfloat162*>(&vw));
            acc[2 * w] += fv.x;
            acc[2 * w + 1] += fv.y;
          } else if constexpr (__is_same(T, fp16)) {
            const float2 fv = __half22float2(
                *reinterpret_cast<const __half2*>(&vw));
            acc[2 * w] += fv.x;
            acc[2 * w + 1] += fv.y;
          } else {
            acc[w] += __uint_as_float(vw);
          }
        }
      }
      T ov[EPU];
#pragma unroll
      for (int q = 0; q < EPU; ++q) fromF(acc[q], ov[q]);
      *reinterpret_cast<u32x4*>(out + (size_t)t * H + h) =
          *reinterpret_cast<const u32x4*>(ov);
    }
  }
}

// static XCD-swizzled walk over one GEMM phase's (e, mT, nT) jobs
// (identical swizzle to the classic persistent grid)
template <typename ET, int PHASE, int BN, int BM, typename WET, int STAGES,
          int CK = 0, int CN = 0>
__device__ __forceinline__ void gemm_phase_walk(const GemmArgs& a, char* smem,
                                                int mT, int nT, int nJobs,
                                                int nBlocks, int act,
                                                int hasBias,
                                                uint32_t* doneCtr,
                                                uint32_t* rowDone,
                                                int debugBits) {
  const int qx = nJobs / 8, rx = nJobs % 8;
  for (int jl = blockIdx.x; jl < nJobs; jl += nBlocks) {
    const int xcd = jl % 8, pos = jl / 8;
    const int swz =
        (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
    const int e = swz / (mT * nT);
    const int rem = swz % (mT * nT);
    const bool ran = gemm_job_body<ET, PHASE, BN, BM, WET, STAGES, CK, CN>(
        a, smem, e, 0, (rem % mT) * BM, (rem / mT) * BN, act, hasBias != 0);
    // job-count arrival: the epilogue's sc1 write-through stores need
    // only a per-wave drain before the relaxed arrival (Guideline 16
    // R1) - no cache-flushing release fence per block. Row arrivals
    // (rowDone) progressively unlock the down phase's dependent tiles.
    if (ran && !(debugBits & 1)) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      __hip_atomic_fetch_add(doneCtr, 1u, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
      if (ran && rowDone)
        __hip_atomic_fetch_add(rowDone + (size_t)e * mT + (rem % mT), 1u,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    }
  }
}

// down-phase walk: each job first waits (bounded) for the up row-tiles
// covering its A rows, takes ONE agent acquire, then runs - the down
// phase overlaps the up phase's tail instead of waiting for all of it
template <typename ET, int BN, int BM, typename WET, int STAGES,
          int CK = 0, int CN = 0>
__device__ __forceinline__ void gemm_phase_walk_dn(
    const GemmArgs& a, const FusedMeta& f, char* smem, int mT, int nT,
    int nJobs, int nBlocks, int hasBias, uint32_t* doneCtr) {
  const int qx = nJobs / 8, rx = nJobs % 8;
  for (int jl = blockIdx.x; jl < nJobs; jl += nBlocks) {
    const int xcd = jl % 8, pos = jl / 8;
    const int swz =
        (xcd < rx ? xcd * (qx + 1) : rx * (qx + 1) + (xcd - rx) * qx) + pos;
    const int e = swz / (mT * nT);
    const int rem = swz % (mT * nT);
    const int m0 = (rem % mT) * BM;
    // routed counts are final (route fan-in acquired earlier)
    const uint32_t rClip =
        min(__hip_atomic_load(f.eCf + e, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT),
            (uint32_t)a.EC);
    bool ran = false;
    if ((uint32_t)m0 < rClip) {
      if (threadIdx.x == 0) {
        // wait for every up row-tile covering rows [m0, min(m0+BM, rClip))
        const int rowHi =
            ((int)rClip - 1 < m0 + BM - 1) ? (int)rClip - 1 : m0 + BM - 1;
        for (int r = m0 / f.upBM; r <= rowHi / f.upBM; ++r) {
          const uint32_t* ctr = f.upRowDone + (size_t)e * f.upMT + r;
          long long spins = 0;
          while (__hip_atomic_load(ctr, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT) <
                 (uint32_t)f.upRowTarget) {
            __builtin_amdgcn_s_sleep(32);
            if (++spins > f.spinBound) {
              if (f.errWord)
                __hip_atomic_store(f.errWord, 3u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
              break;
            }
          }
        }
        if (!(f.debugBits & 2))
          __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
      }
      __syncthreads();
      ran = gemm_job_body<ET, 1, BN, BM, WET, STAGES, CK, CN>(
          a, smem, e, 0, m0, (rem / mT) * BN, 0, hasBias != 0);
    }
    if (ran && !(f.debugBits & 1)) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
    if (threadIdx.x == 0)
      __hip_atomic_fetch_add(doneCtr, 1u, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
  }
}

template <typename ET, typename WET, int CH = 0, int CP = 0>
__global__ __launch_bounds__(512, 1) void k_moe_fused(GemmArgs gl, GemmArgs gu,
                                                      GemmArgs gd,
                                                      FusedMeta f) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint32_t* sCtl = reinterpret_cast<uint32_t*>(smem + f.ctlOff);
  const int tid = threadIdx.x;
  if (tid == 0)
    atomicMax(reinterpret_cast<unsigned long long*>(&f.ctl->clk[0]),
              __builtin_amdgcn_s_memrealtime());

  // ---- phase L: gate logits tiles + inline route on last arrival ----
  const int jobsPerTile = f.lgKS * f.lgNT;
  for (int j = blockIdx.x; j < f.lgJobs; j += f.nBlocks) {
    const int t = j % f.nTiles;
    const int rest = j / f.nTiles;
    const int ks = rest % f.lgKS;
    const int nt = rest / f.lgKS;
    (void)gemm_job_body<ET, 3, 128, 128, ET, 2, CH, 0>(gl, smem, 0, ks,
                                                        t * 128, nt * 128, 0,
                                                        false);
    // arrive on the tile: the logits writes are fp32 atomicAdds
    // (globally coherent), so only a vmcnt drain orders them before
    // the arrival; no cache release is needed
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (tid == 0)
      *sCtl = __hip_atomic_fetch_add(f.tileArrive + t, 1u, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();
    const bool last = (*sCtl == (uint32_t)(jobsPerTile - 1));
    __syncthreads();  // everyone read sCtl before the next iteration
    if (last) {
      if (tid == 0) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
      __syncthreads();
      switch (f.topk) {
        case 1: route_tile<ET, 1>(smem, f, t); break;
        case 2: route_tile<ET, 2>(smem, f, t); break;
        case 3: route_tile<ET, 3>(smem, f, t); break;
        case 4: route_tile<ET, 4>(smem, f, t); break;
        case 5: route_tile<ET, 5>(smem, f, t); break;
        case 6: route_tile<ET, 6>(smem, f, t); break;
        case 7: route_tile<ET, 7>(smem, f, t); break;
        default: route_tile<ET, 8>(smem, f, t); break;
      }
      fused_release_arrive(&f.ctl->routeTilesDone);
    }
  }
  fused_wait(&f.ctl->routeTilesDone, (uint32_t)f.nTiles, f.errWord,
             f.spinBound, (f.debugBits & 2) != 0);
  if (tid == 0)
    atomicMax(reinterpret_cast<unsigned long long*>(&f.ctl->clk[1]),
              __builtin_amdgcn_s_memrealtime());

  // ---- phase U: expert up-GEMM ----
  switch (f.upGeom) {
    case 0:
      gemm_phase_walk<ET, 0, 256, 256, WET, 2, CH, CP>(
          gu, smem, f.upMT, f.upNT, f.upJobs, f.nBlocks, gu.act,
          gu.bias != nullptr, &f.ctl->upJobsDone, f.upRowDone,
          f.debugBits);
      break;
    case 1:
      gemm_phase_walk<ET, 0, 128, 256, WET, 3, CH, CP>(
          gu, smem, f.upMT, f.upNT, f.upJobs, f.nBlocks, gu.act,
          gu.bias != nullptr, &f.ctl->upJobsDone, f.upRowDone,
          f.debugBits);
      break;
    case 3:
      gemm_phase_walk<ET, 0, 256, 128, WET, 3, CH, CP>(
          gu, smem, f.upMT, f.upNT, f.upJobs, f.nBlocks, gu.act,
          gu.bias != nullptr, &f.ctl->upJobsDone, f.upRowDone,
          f.debugBits);
      break;
    default:
      gemm_phase_walk<ET, 0, 128, 128, WET, 2, CH, CP>(
          gu, smem, f.upMT, f.upNT, f.upJobs, f.nBlocks, gu.act,
          gu.bias != nullptr, &f.ctl->upJobsDone, f.upRowDone,
          f.debugBits);
      break;
  }
  // no global up->down seam: each down job waits only for the up
  // row-tiles covering its A rows (gemm_phase_walk_dn); clk[2] marks
  // this block's transition into the down walk
  if (tid == 0)
    atomicMax(reinterpret_cast<unsigned long long*>(&f.ctl->clk[2]),
              __builtin_amdgcn_s_memrealtime());

  // ---- phase D: expert down-GEMM with combine-slot epilogue ----
  switch (f.dnGeom) {
    case 0:
      gemm_phase_walk_dn<ET, 256, 256, WET, 2, CP, CH>(
          gd, f, smem, f.dnMT, f.dnNT, f.dnJobs, f.nBlocks,
          gd.bias != nullptr, &f.ctl->dnJobsDone);
      break;
    case 1:
      gemm_phase_walk_dn<ET, 128, 256, WET, 3, CP, CH>(
          gd, f, smem, f.dnMT, f.dnNT, f.dnJobs, f.nBlocks,
          gd.bias != nullptr, &f.ctl->dnJobsDone);
      break;
    case 3:
      gemm_phase_walk_dn<ET, 256, 128, WET, 3, CP, CH>(
          gd, f, smem, f.dnMT, f.dnNT, f.dnJobs, f.nBlocks,
          gd.bias != nullptr, &f.ctl->dnJobsDone);
      break;
    default:
      gemm_phase_walk_dn<ET, 128, 128, WET, 2, CP, CH>(
          gd, f, smem, f.dnMT, f.dnNT, f.dnJobs, f.nBlocks,
          gd.bias != nullptr, &f.ctl->dnJobsDone);
      break;
  }
  fused_wait(&f.ctl->dnJobsDone, (uint32_t)f.dnJobs, f.errWord,
             f.spinBound, (f.debugBits & 2) != 0);
  if (tid == 0)
    atomicMax(reinterpret_cast<unsigned long long*>(&f.ctl->clk[3]),
              __builtin_amdgcn_s_memrealtime());

  // ---- phase C: masked slot reduction ----
  for (int j = blockIdx.x; j < f.cbJobs; j += f.nBlocks) {
    const int t0 = j * f.cbTB;
    const int t1 = min(f.S, t0 + f.cbTB);
    switch (f.topk) {
      case 1: combine_tokens<ET, 1>(f, t0, t1); break;
      case 2: combine_tokens<ET, 2>(f, t0, t1); break;
      case 3: combine_tokens<ET, 3>(f, t0, t1); break;
      case 4: combine_tokens<ET, 4>(f, t0, t1); break;
      case 5: combine_tokens<ET, 5>(f, t0, t1); break;
      case 6: combine_tokens<ET, 6>(f, t0, t1); break;
      case 7: combine_tokens<ET, 7>(f, t0, t1); break;
      default: combine_tokens<ET, 8>(f, t0, t1); break;
    }
  }
  if (tid == 0)
    atomicMax(reinterpret_cast<unsigned long long*>(&f.ctl->clk[4]),
              __builtin_amdgcn_s_memrealtime());
}

// ===========================================================================
// Host side: state, workspace, ABI
// ===========================================================================

static int launch_group_gemm(hipStream_t st, int phase, const struct GemmArgs& a,
                             int M, int nE);

namespace {

struct State {
  fm_config cfg{};
  bool initialized = false;
  int rank = 0, world = 1;
  int S = 0, H = 0, P = 0, E = 0, PX = 0, nLx = 0, EC = 0, pEC = 0;
  size_t esz = 0;
  size_t wesz = 0;  // expert-weight element size (1 for fp8 weights)
  // workspace
  TPS* tokenIds = nullptr;   // [E, pEC]
  uint32_t* eC = nullptr;    // [E]
  float* logits32 = nullptr; // [S, E] gate logit accumulator
  float* gML = nullptr;      // [E] training aux: mean gate prob
  float* gMeC = nullptr;     // [E] training aux: routed fraction
  // one-sided P2P heap (opt-in EP transport)
  char* heap = nullptr;      // local block: recv | ret | dispFlags | retFlags
  size_t heapRecvOff = 0, heapRetOff = 0, heapDFlagOff = 0, heapRFlagOff = 0;
  size_t heapBytes = 0;
  void* peerBase[64] = {};   // opened peer heap bases (self = heap)
  bool peerOpened[64] = {};  // true only for real hipIpcOpenMemHandle maps
  uint64_t* dPeerRecv = nullptr;   // device tables [world]
  uint64_t* dPeerRet = nullptr;
  uint64_t* dPeerDFlag = nullptr;
  uint64_t* dPeerRFlag = nullptr;
  uint32_t* dArrive = nullptr;     // [2*E] dispatch/return arrival counters
  uint32_t* hP2pErr = nullptr;     // host-mapped timeout flag (zero-copy)
  uint32_t* dP2pErr = nullptr;     // device view of hP2pErr
  unsigned long long seq = 0;
  // transparent hipGraph cache for repeated-pointer forwards
  hipGraphExec_t graphExec = nullptr;
  void* graphKey[8] = {};
  void* lastKey[8] = {};
  bool graphValid = false;
  void* xM = nullptr;        // [nLx_alloc, pEC, P] Element
  uint8_t* x8 = nullptr;     // dtype 5: quantized activations [S, H]
  uint8_t* xs = nullptr;     // dtype 5: x block scales [S, H/64]
  uint8_t* xM8 = nullptr;    // dtype 5: quantized intermediates
  uint8_t* xMs = nullptr;    // dtype 5: xM block scales
  float* O32 = nullptr;      // [S, H] (staged-API combine accumulator)
  void* cbuf = nullptr;      // [S, k, H] Element: non-atomic combine slots
  uint8_t* kept = nullptr;   // [S, k] capacity-kept mask (gate-written)
  int nLxAlloc = 0;
  // fused persistent kernel (single-launch forward)
  FusedCtl* fusedCtl = nullptr;   // device: FusedCtl + tileArrive[nTiles]
  size_t fusedCtlBytes = 0;
  uint32_t* hFusedErr = nullptr;  // host-mapped give-up flag (zero-copy)
  uint32_t* dFusedErr = nullptr;
  int nCU = 0;
  int wallClockKHz = 100000;      // s_memrealtime rate (ticks per ms)
  bool lastForwardFused = false;
};
State g;

thread_local char g_err[512] = "";

void setErr(const char* msg) {
  snprintf(g_err, sizeof(g_err), "%s", msg);
}

#define FM_HIP_CHECK(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      snprintf(g_err, sizeof(g_err), "HIP error %s at %s:%d: %s",            \
               hipGetErrorString(_e), __FILE__, __LINE__, #expr);            \
      return FM_ERR_HIP;                                                     \
    }                                                                        \
  } while (0)

// dtype 5: does the MX grouped GEMM pick the 256^2 tile for (M, N)?
// (mirrored by launch_group_gemm's MX branch; the 256-tile up epilogue
// quantizes in-register, making the xM quant pass unnecessary)
static bool mxBigGeom(int M, int N, int nE) {
  return (M >= 256) && (N >= 256) &&
         DIVUP(M, 256) * DIVUP(N, 256) * nE >= (g.nCU > 0 ? g.nCU : 256);
}

// 256-wide-N MX geometry (128x256 default or FM_MX_GEOM=big): the
// epilogue quantizes in-register in both, so the xM quant pass is
// skipped whenever the up GEMM runs a 256-wide tile
static bool mxWideN(int M, int N, int nE) {
  return (N >= 256) &&
         (mxBigGeom(M, N, nE) ||
          DIVUP(M, 128) * DIVUP(N, 256) * nE >= (g.nCU > 0 ? g.nCU : 256));
}

size_t gate_lds_bytes(int E, size_t esz) {
  // mirrors the k_gate arena: logits + A chunk + G chunk
  return 128 * (E + 1) * sizeof(float) + (128 + E) * (64 + 8) * esz;
}

int launch_gate(hipStream_t st, const void* x, const void* gate_w,
                void* gate_out, int64_t S) {
  // split gate: logits GEMM parallelised over (token tile, H chunk),
  // then per-tile softmax/top-k/route (see k_gate_logits/k_gate_route)
  const int tiles = (int)(S / 128);
  int chunksWanted = DIVUP(512, tiles);
  int Hc = DIVUP(DIVUP(g.H, chunksWanted), 64) * 64;
  const int chunks = DIVUP(g.H, Hc);
  const int eChunks = DIVUP(g.E, 128);
  const int Ec = g.E < 128 ? g.E : 128;
  const bool mfmaLogits = (g.esz == 2);  // bf16/fp16: MFMA logits GEMM
  const size_t ldsL = gate_lds_bytes(Ec, g.esz);
  // route arena: TT = 64-token half-passes at E > 256 (route_tile_core)
  const int rTT = (g.E > 256) ? 64 : 128;
  const size_t ldsR = (size_t)rTT * (g.E + 1) * sizeof(float) +
                      rTT * 8 * 2 * sizeof(uint16_t) + g.E * sizeof(uint32_t) +
                      3 * rTT * sizeof(float) + 64;
  float* gMLp = g.cfg.is_training ? g.gML : nullptr;
  float* gMeCp = g.cfg.is_training ? g.gMeC : nullptr;
  if (g.cfg.is_training)
    FM_HIP_CHECK(hipMemsetAsync(g.gML, 0, 2 * (size_t)g.E * sizeof(float), st));
#define GATE_LOGITS(T)                                                        \
  do {                                                                        \
    if (ldsL > 64 * 1024)                                                     \
      (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&k_gate_logits<T>),   \
                          hipFuncAttributeMaxDynamicSharedMemorySize,         \
                          (int)ldsL);                                         \
    hipLaunchKernelGGL((k_gate_logits<T>), dim3(tiles, chunks, eChunks),     \
                       dim3(256), ldsL, st, reinterpret_cast<const T*>(x),    \
                       reinterpret_cast<const T*>(gate_w), g.logits32, g.eC,  \
                       (int)S, g.H, g.E, Hc);                                 \
  } while (0)
#define GATE_ROUTE(T, KK)                                                     \
  do {                                                                        \
    if (ldsR > 64 * 1024)                                                     \
      (void)hipFuncSetAttribute(                                                    \
          reinterpret_cast<const void*>(&k_gate_route<T, KK>),                \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)ldsR);             \
    hipLaunchKernelGGL((k_gate_route<T, KK>), dim3(tiles), dim3(512), ldsR,   \
                       st, g.logits32, reinterpret_cast<T*>(gate_out),        \
                       g.tokenIds, g.eC, (int)S, g.E, g.PX, g.EC, g.pEC,      \
                       gMLp, gMeCp, g.kept);                                  \
  } while (0)
#define GATE_K(T)                                                             \
  switch (g.cfg.expert_top_k) {                                               \
    case 1: GATE_ROUTE(T, 1); break;                                          \
    case 2: GATE_ROUTE(T, 2); break;                                          \
    case 3: GATE_ROUTE(T, 3); break;                                          \
    case 4: GATE_ROUTE(T, 4); break;                                          \
    case 5: GATE_ROUTE(T, 5); break;                                          \
    case 6: GATE_ROUTE(T, 6); break;                                          \
    case 7: GATE_ROUTE(T, 7); break;                                          \
    case 8: GATE_ROUTE(T, 8); break;                                          \
    default: setErr("unsupported expert_top_k (1..8)"); return FM_ERR_UNSUPPORTED; \
  }
  if (mfmaLogits) {
    GemmArgs ga{};
    ga.A = x;
    ga.B = gate_w;
    ga.out = g.logits32;
    ga.tokenIds = nullptr;
    ga.eC = g.eC;      // PHASE 3 zeroes it (block 0) for the route kernel
    ga.EC = g.E;
    ga.K = g.H;
    ga.N = g.E;
    ga.nRows = (int)S;
    ga.H = g.H;
    ga.splitK = 1;
    int rc = launch_group_gemm(st, 3, ga, (int)S, 1);
    if (rc != FM_OK) return rc;
  }
  if (g.cfg.dtype == 2 || g.cfg.dtype == 4 || g.cfg.dtype == 5) {
    if (!mfmaLogits) GATE_LOGITS(bf16);
    GATE_K(bf16)
  } else if (g.cfg.dtype == 3) {
    if (!mfmaLogits) GATE_LOGITS(fp16);
    GATE_K(fp16)
  } else {
    GATE_LOGITS(float);
    GATE_K(float)
  }
#undef GATE_K
#undef GATE_ROUTE
#undef GATE_LOGITS
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

// ---------------------------------------------------------------------------
// Fused single-launch forward: geometry selection + launch (device side
// above, k_moe_fused). Returns FM_FALLBACK when the fused kernel cannot
// run this shape (classic multi-kernel path takes over).
// ---------------------------------------------------------------------------

static const int FM_FALLBACK = 1000;  // internal sentinel, never ABI-visible

static bool fusedEnabled() {
  // FM_FUSED=1 forces the single-launch kernel, FM_FUSED=0 forces the
  // classic multi-kernel path. Default: fused where it measures faster
  // (deep-K shapes where the classic down-GEMM cannot run its
  // 2-blocks/CU 128^2 mode, P > 8192: cfg4-shape +3%); classic
  // otherwise (cfg2/3/5 measured faster multi-kernel this round -
  // profiles/r02; the fused kernel remains the FM_FUSED=1 path under
  // active optimization).
  // re-read per call: tests toggle it to force fused-path coverage
  const char* e = getenv("FM_FUSED");
  if (e) return e[0] != '0';
  return g.P > 8192;
}

static long long fusedSpinBound() {
  static const long long b = [] {
    const char* e = getenv("FM_FUSED_SPIN_LOG2");
    const int lg = e ? atoi(e) : 24;
    return 1ll << (lg < 4 ? 4 : (lg > 40 ? 40 : lg));
  }();
  return b;
}

// ctl-embedded eC mirror (the fused kernel's routed counts; g.eC is
// untouched by the fused path)
static uint32_t* fusedECPtr() {
  return reinterpret_cast<uint32_t*>(g.fusedCtl + 1) + (g.S / 128) +
         (size_t)g.E * (g.pEC / 128);
}

static int fusedPoisonCheck() {
  if (g.hFusedErr &&
      *reinterpret_cast<volatile uint32_t*>(g.hFusedErr)) {
    *g.hFusedErr = 0;  // ack so the caller may retry
    setErr("fused forward gave up on an in-kernel wait (output poisoned)");
    return FM_ERR_HIP;
  }
  return FM_OK;
}

// host mirror of gemm_lds_bytes<> for the dynamic arena
static int gemmLdsBytesRT(int BM, int BN, int BEZ, int stages) {
  const int SE = (stages == 3 && 3 * BM * 64 * 2 + 3 * BN * 64 * BEZ +
                                      BM * 8 + 16 > 160 * 1024)
                     ? 2 : stages;
  return SE * BM * 64 * 2 + SE * BN * 64 * BEZ + BM * 8 + 16;
}

static void fusedGeoDims(int gm, int& bm, int& bn, int& stages) {
  bm = (gm == 0 || gm == 1) ? 256 : 128;
  bn = (gm == 0 || gm == 3) ? 256 : 128;
  stages = (gm == 1 || gm == 3) ? 3 : 2;
}

struct FusedGeo { int geom, mt, nt, jobs; };

// pick the largest tile that still fills the resident grid (mirrors the
// classic mode selection, DESIGN.md par.3, restricted to 1 block/CU)
static FusedGeo fusedPickGeo(int M, int N, int E, int nBlocks,
                             const int* order, int nOrder, int forceGeom) {
  FusedGeo last{};
  for (int i = 0; i < nOrder; ++i) {
    const int gm = order[i];
    int bm, bn, st;
    fusedGeoDims(gm, bm, bn, st);
    if (forceGeom >= 0 && gm != forceGeom) continue;
    if (M < bm && bm > 128) continue;  // do not waste >half the M rows
    const int mt = DIVUP(M, bm), nt = DIVUP(N, bn);
    last = FusedGeo{gm, mt, nt, mt * nt * E};
    if (forceGeom >= 0) return last;
    if (N < bn && bn > 128) continue;  // half-empty B tiles: next option
    if (last.jobs >= nBlocks) return last;
  }
  const int mt = DIVUP(M, 128), nt = DIVUP(N, 128);
  return FusedGeo{4, mt, nt, mt * nt * E};
}

// co-resident blocks/CU for the fused kernel at a given arena size
// (0 = cannot run). The grid is sized to exactly occ x nCU so every
// block is resident and the in-kernel fan-in waits cannot deadlock.
template <typename ET, typename WET, int CH, int CP>
static int fusedOccT(int arena) {
  static int maxDyn = -1;
  if (maxDyn < 0) {
    maxDyn = (hipFuncSetAttribute(
                  reinterpret_cast<const void*>(&k_moe_fused<ET, WET, CH, CP>),
                  hipFuncAttributeMaxDynamicSharedMemorySize,
                  160 * 1024) == hipSuccess)
                 ? 160 * 1024 : 64 * 1024;
  }
  if (arena > maxDyn) return 0;
  static int occArena = -1, occVal = 0;
  if (occArena != arena) {
    occVal = 0;
    if (hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &occVal, k_moe_fused<ET, WET, CH, CP>, 512, arena) != hipSuccess)
      occVal = 0;
    occArena = arena;
  }
  return occVal;
}

template <typename ET, typename WET>
static int launchFusedT(hipStream_t st, const GemmArgs& gl,
                        const GemmArgs& gu, const GemmArgs& gd,
                        const FusedMeta& f, int arena, int nBlocks) {
  // BASELINE shape dictionary: compile-time (H, P) fold the tile loop's
  // address math and trip counts (the reference freezes these in its
  // compile-time config contract too); generic fallback otherwise
#define FM_LAUNCH_SHAPE(HH, PP)                                                 do {                                                                            hipLaunchKernelGGL((k_moe_fused<ET, WET, HH, PP>), dim3(nBlocks),                                dim3(512), arena, st, gl, gu, gd, f);                      FM_HIP_CHECK(hipGetLastError());                                              return FM_OK;                                                               } while (0)
  if (g.H == 1024 && g.P == 4096) FM_LAUNCH_SHAPE(1024, 4096);
  if (g.H == 2048 && g.P == 8192) FM_LAUNCH_SHAPE(2048, 8192);
  if (g.H == 4096 && g.P == 14336) FM_LAUNCH_SHAPE(4096, 14336);
  FM_LAUNCH_SHAPE(0, 0);
#undef FM_LAUNCH_SHAPE
}

template <typename ET, typename WET>
static int fusedOccShape(int arena) {
  if (g.H == 1024 && g.P == 4096) return fusedOccT<ET, WET, 1024, 4096>(arena);
  if (g.H == 2048 && g.P == 8192) return fusedOccT<ET, WET, 2048, 8192>(arena);
  if (g.H == 4096 && g.P == 14336)
    return fusedOccT<ET, WET, 4096, 14336>(arena);
  return fusedOccT<ET, WET, 0, 0>(arena);
}

static int fusedOcc(int arena) {
  if (g.cfg.dtype == 3) return fusedOccShape<fp16, fp16>(arena);
  if (g.cfg.dtype == 4) return fusedOccShape<bf16, fp8e4m3>(arena);
  return fusedOccShape<bf16, bf16>(arena);
}

template <typename ET, typename WET>
static int launchFused(hipStream_t st, const GemmArgs& gl,
                       const GemmArgs& gu, const GemmArgs& gd,
                       const FusedMeta& f, int arena, int nBlocks) {
  return launchFusedT<ET, WET>(st, gl, gu, gd, f, arena, nBlocks);
}

static int moe_forward_fused(hipStream_t st, const void* x, const void* gate_w,
                             const void* expert_w, const void* b_up,
                             const void* b_dn, void* gate_out,
                             void* moe_out) {
  if (!g.fusedCtl || g.nCU <= 0) return FM_FALLBACK;
  int nBlocks = g.nCU;  // refined to occ x nCU once the arena is known
  const int K = g.cfg.expert_top_k;
  const int BEZ = (int)g.wesz;

  // logits jobs: 128x128 K-split tiles, enough to fill the grid
  const int nTiles = g.S / 128;
  const int lgNT = DIVUP(g.PX, 128);
  int lgKS = 1;
  while (nTiles * lgNT * lgKS < nBlocks && lgKS < 8 &&
         (g.H / 64) % (lgKS * 2) == 0)
    lgKS *= 2;
  const int lgJobs = nTiles * lgNT * lgKS;

  static const int forceUp = [] {
    const char* e = getenv("FM_FUSED_UP");
    return e ? atoi(e) : -1;
  }();
  static const int forceDn = [] {
    const char* e = getenv("FM_FUSED_DN");
    return e ? atoi(e) : -1;
  }();
  const int upOrder[3] = {0, 3, 4};
  const int dnOrder[3] = {1, 3, 4};
  const FusedGeo up = fusedPickGeo(g.pEC, g.P, g.E, nBlocks, upOrder, 3,
                                   forceUp);
  const FusedGeo dn = fusedPickGeo(g.pEC, g.H, g.E, nBlocks, dnOrder, 3,
                                   forceDn);

  // LDS arena: max of every tile body used plus the route scratch,
  // plus the 16-B control slice at the tail
  int bm, bn, stg;
  int arena = gemmLdsBytesRT(128, 128, (int)g.esz, 2);  // logits tile
  fusedGeoDims(up.geom, bm, bn, stg);
  arena = std::max(arena, gemmLdsBytesRT(bm, bn, BEZ, stg));
  fusedGeoDims(dn.geom, bm, bn, stg);
  arena = std::max(arena, gemmLdsBytesRT(bm, bn, BEZ, stg));
  const int rTT = (g.E > 256) ? 64 : 128;  // route half-passes at big E
  const int routeBytes = rTT * (g.E + 1) * 4 + rTT * K * 2 +
                         (rTT * K + 2) * 2 + g.E * 4 + 3 * rTT * 4;
  arena = std::max(arena, routeBytes);
  arena = (arena + 15) / 16 * 16 + 16;
  if (arena > 160 * 1024) return FM_FALLBACK;
  const int occ = fusedOcc(arena);
  if (occ < 1) return FM_FALLBACK;
  nBlocks = std::min(occ, 2) * g.nCU;  // all blocks co-resident

  GemmArgs gl{}, gu{}, gd{};
  FusedMeta f{};
  gl.A = x;
  gl.B = gate_w;
  gl.out = g.logits32;
  gl.K = g.H;
  gl.N = g.E;
  gl.nRows = g.S;
  gl.H = g.H;
  gl.splitK = lgKS;
  // always atomic: the route runs in a DIFFERENT block than the logits
  // tiles, and atomics are globally coherent without a release fence
  gl.atomicLogits = 1;

  gu.A = x;
  gu.B = expert_w;
  gu.bias = b_up;
  gu.out = g.xM;
  gu.gate_out = gate_out;
  gu.tokenIds = g.tokenIds;
  gu.eC = nullptr;  // set to the ctl-embedded eC mirror below
  gu.strideAExpert = 0;
  gu.strideBExpert = 2LL * g.P * g.H;
  gu.strideOExpert = (long long)g.pEC * g.P;
  gu.K = g.H;
  gu.N = g.P;
  gu.EC = g.EC;
  gu.pEC = g.pEC;
  gu.PX = g.PX;
  gu.topk = K;
  gu.act = g.cfg.hidden_act;
  gu.H = g.H;
  gu.splitK = 1;
  static const int dbg = [] {
    const char* e = getenv("FM_FUSED_DEBUG");
    return e ? atoi(e) : 0;
  }();
  gu.sc1Out = (dbg & 4) ? 0 : 1;  // xM consumed by down-GEMM blocks

  gd = gu;
  gd.A = g.xM;
  gd.B = reinterpret_cast<const char*>(expert_w) +
           (size_t)g.P * g.H * g.wesz;
  gd.bias = b_dn;
  gd.out = nullptr;
  gd.O32 = reinterpret_cast<float*>(g.cbuf);
  gd.moe_out = moe_out;
  gd.strideAExpert = (long long)g.pEC * g.P;
  gd.K = g.P;
  gd.N = g.H;
  gd.act = 0;
  gd.slotAlways = 1;

  f.ctl = g.fusedCtl;
  f.tileArrive = reinterpret_cast<uint32_t*>(g.fusedCtl + 1);
  f.upRowDone = f.tileArrive + nTiles;
  fusedGeoDims(up.geom, bm, bn, stg);
  f.upBM = bm;
  f.upRowTarget = up.nt;
  f.eCf = fusedECPtr();  // ctl-embedded (one memset covers it)
  f.logits32 = g.logits32;
  f.gate_out = gate_out;
  f.tokenIds = g.tokenIds;
  f.cbuf = g.cbuf;
  f.kept = g.kept;
  f.moe_out = moe_out;
  f.gML = g.cfg.is_training ? g.gML : nullptr;
  f.gMeC = g.cfg.is_training ? g.gMeC : nullptr;
  f.errWord = g.dFusedErr;
  f.S = g.S;
  f.H = g.H;
  f.E = g.E;
  f.PX = g.PX;
  f.EC = g.EC;
  f.pEC = g.pEC;
  f.topk = K;
  f.training = g.cfg.is_training;
  f.nTiles = nTiles;
  f.lgKS = lgKS;
  f.lgNT = lgNT;
  f.lgJobs = lgJobs;
  f.upGeom = up.geom;
  f.upMT = up.mt;
  f.upNT = up.nt;
  f.upJobs = up.jobs;
  f.dnGeom = dn.geom;
  f.dnMT = dn.mt;
  f.dnNT = dn.nt;
  f.dnJobs = dn.jobs;
  f.cbTB = 8;
  f.cbJobs = DIVUP(g.S, 8);
  f.debugBits = dbg;
  f.nBlocks = nBlocks;
  f.ctlOff = arena - 16;
  f.spinBound = fusedSpinBound();

  // per-forward state re-init (Guideline 16: zero every polled word
  // ahead of the launch; these become memset nodes under graph capture)
  gu.eC = f.eCf;
  gd.eC = f.eCf;
  FM_HIP_CHECK(hipMemsetAsync(g.fusedCtl, 0, g.fusedCtlBytes, st));
  if (g.cfg.is_training)
    FM_HIP_CHECK(hipMemsetAsync(g.gML, 0, 2 * (size_t)g.E * sizeof(float), st));

  if (g.cfg.dtype == 3)
    return launchFused<fp16, fp16>(st, gl, gu, gd, f, arena, nBlocks);
  if (g.cfg.dtype == 4)
    return launchFused<bf16, fp8e4m3>(st, gl, gu, gd, f, arena, nBlocks);
  return launchFused<bf16, bf16>(st, gl, gu, gd, f, arena, nBlocks);
}


}  // namespace

extern "C" {

const char* fm_last_error(void) { return g_err; }
int fm_built_for_gfx950(void) { return 1; }

int fm_initialize(const fm_config* cfg, int rank, int world_size) {
  if (g.initialized) { setErr("already initialized"); return FM_ERR_STATE; }
  if (!cfg) { setErr("null config"); return FM_ERR_STATE; }
  g.cfg = *cfg;
  g.rank = rank;
  g.world = world_size;
  g.S = cfg->sequence_len * cfg->mini_batch;
  g.H = cfg->hidden_size;
  g.P = cfg->intermediate_size;
  g.E = cfg->num_experts;
  g.PX = DIVUP(g.E, 64) * 64;
  if (cfg->num_experts % world_size != 0) {
    setErr("num_experts must divide by world_size (uniform EP split)");
    return FM_ERR_SHAPE;
  }
  g.nLx = g.E / world_size;
  const int base = cfg->drop_tokens ? DIVUP(g.S, g.E) : g.S;
  g.EC = base * cfg->capacity_factor * cfg->expert_top_k;
  g.pEC = DIVUP(g.EC, 128) * 128;
  switch (cfg->dtype) {
    case 0: case 1: g.esz = 4; break;
    case 2: case 3: g.esz = 2; break;
    case 4: g.esz = 2; break;  // fp8e4m3 expert weights, bf16 activations
    case 5: g.esz = 2; break;  // MX: fp8 weights + runtime-quantized
                               // fp8 activations on the scaled MFMA
    default: setErr("unknown dtype");
             return FM_ERR_UNSUPPORTED;
  }
  g.wesz = (cfg->dtype == 4 || cfg->dtype == 5) ? 1 : g.esz;
  if (cfg->dtype == 5 && (cfg->hidden_size % 128 || cfg->intermediate_size % 128)) {
    setErr("dtype 5 (MX fp8) requires H, P multiples of 128 (BK=128)");
    return FM_ERR_SHAPE;
  }
  if (g.E > 512) { setErr("E > 512 not supported this round"); return FM_ERR_UNSUPPORTED; }
  if (world_size > 64) {
    // the P2P peer tables (State.peerBase, fm_heap_connect locals) are
    // sized for 64 ranks - a full 8-node envelope; reject beyond it
    setErr("world_size > 64 not supported");
    return FM_ERR_UNSUPPORTED;
  }
  if (g.S < 128 || g.S % 128) {
    // the gate tiles tokens in blocks of 128 (launch_gate: tiles = S/128);
    // a ragged tail would be silently dropped - reject instead
    setErr("S = sequence_len * mini_batch must be a positive multiple of 128");
    return FM_ERR_SHAPE;
  }
  if (cfg->expert_top_k < 1 || cfg->expert_top_k > 8 ||
      cfg->expert_top_k > g.E) {
    setErr("expert_top_k must be in [1, min(E, 8)]");
    return FM_ERR_UNSUPPORTED;
  }
  if (g.H % 64 || g.P % 64) { setErr("H and P must be multiples of 64"); return FM_ERR_SHAPE; }
  // (bf16/fp16 tiles handle H, P % 64 via the B-row clamp and col < N
  // epilogue guards - the round-1 %128 restriction is lifted, matching
  // the schema's contract; covered by the H=192/P=320 parity cases)
  // workspace: tokenIds/eC global-E; xM sized for the worst consumer
  // (single-rank path: E experts; EP path: world*EC rows per local expert)
  g.nLxAlloc = (world_size == 1) ? g.E
                                 : g.nLx * DIVUP(world_size * g.EC, g.pEC);
  FM_HIP_CHECK(hipMalloc(&g.tokenIds, (size_t)g.E * g.pEC * sizeof(TPS)));
  FM_HIP_CHECK(hipMalloc(&g.logits32, (size_t)g.S * g.E * sizeof(float)));
  FM_HIP_CHECK(hipMemset(g.logits32, 0, (size_t)g.S * g.E * sizeof(float)));
  FM_HIP_CHECK(hipMalloc(&g.gML, 2 * (size_t)g.E * sizeof(float)));
  g.gMeC = g.gML + g.E;
  FM_HIP_CHECK(hipMalloc(&g.eC, (size_t)g.E * sizeof(uint32_t)));
  FM_HIP_CHECK(hipMalloc(&g.xM, (size_t)g.nLxAlloc * g.pEC * g.P * g.esz));
  FM_HIP_CHECK(hipMalloc(&g.O32, (size_t)g.S * g.H * sizeof(float)));
  FM_HIP_CHECK(hipMemset(g.O32, 0, (size_t)g.S * g.H * sizeof(float)));
  FM_HIP_CHECK(hipMalloc(&g.cbuf,
                         (size_t)g.S * g.cfg.expert_top_k * g.H * g.esz));
  FM_HIP_CHECK(hipMalloc(&g.kept, (size_t)g.S * g.cfg.expert_top_k));
  if (g.cfg.dtype == 5) {
    // sized for BOTH the single-rank forward (S rows) and the EP
    // segments path (up to nLxAlloc*pEC recv rows)
    const size_t qRows = std::max((size_t)g.S, (size_t)g.nLxAlloc * g.pEC);
    FM_HIP_CHECK(hipMalloc(&g.x8, qRows * g.H));
    FM_HIP_CHECK(hipMalloc(&g.xs, qRows * (g.H / 64)));
    FM_HIP_CHECK(hipMalloc(&g.xM8, (size_t)g.nLxAlloc * g.pEC * g.P));
    FM_HIP_CHECK(hipMalloc(&g.xMs,
                           (size_t)g.nLxAlloc * g.pEC * (g.P / 64)));
  }
  if (g.esz == 2) {
    // fused persistent kernel state: control block + per-tile arrival
    // counters (zeroed per forward by a memset node) + host-mapped
    // give-up flag
    const int nTiles = g.S / 128;
    // + per-tile logits arrivals + per-(expert, row-tile) up completion
    // counters (worst case BM=128 row tiles) + an embedded eC mirror so
    // ONE memset node re-initializes every per-forward word (a second
    // 32-B memset blit costs ~4 us of wall per step)
    g.fusedCtlBytes = sizeof(FusedCtl) + (size_t)nTiles * sizeof(uint32_t) +
                      (size_t)g.E * (g.pEC / 128) * sizeof(uint32_t) +
                      (size_t)g.E * sizeof(uint32_t);
    FM_HIP_CHECK(hipMalloc(&g.fusedCtl, g.fusedCtlBytes));
    FM_HIP_CHECK(hipMemset(g.fusedCtl, 0, g.fusedCtlBytes));
    FM_HIP_CHECK(hipHostMalloc(&g.hFusedErr, sizeof(uint32_t),
                               hipHostMallocMapped));
    *g.hFusedErr = 0;
    FM_HIP_CHECK(hipHostGetDevicePointer(
        reinterpret_cast<void**>(&g.dFusedErr), g.hFusedErr, 0));
    int dev = 0;
    FM_HIP_CHECK(hipGetDevice(&dev));
    int ncu = 0;
    FM_HIP_CHECK(hipDeviceGetAttribute(
        &ncu, hipDeviceAttributeMultiprocessorCount, dev));
    g.nCU = ncu;
    int khz = 0;
    if (hipDeviceGetAttribute(&khz, hipDeviceAttributeWallClockRate, dev) ==
            hipSuccess && khz > 0)
      g.wallClockKHz = khz;
  }
  g.initialized = true;
  return FM_OK;
}

int fm_finalize(void) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (g.graphExec) { (void)hipGraphExecDestroy(g.graphExec); g.graphExec = nullptr; }
  (void)hipFree(g.tokenIds); (void)hipFree(g.eC); (void)hipFree(g.xM);
  (void)hipFree(g.O32);
  (void)hipFree(g.cbuf); (void)hipFree(g.kept);
  (void)hipFree(g.logits32); (void)hipFree(g.gML);
  if (g.x8) { (void)hipFree(g.x8); (void)hipFree(g.xs);
              (void)hipFree(g.xM8); (void)hipFree(g.xMs); }
  if (g.fusedCtl) (void)hipFree(g.fusedCtl);
  if (g.hFusedErr) (void)hipHostFree(g.hFusedErr);
  if (g.heap) {
    for (int p = 0; p < g.world; ++p) {
      // close only real IPC mappings: a NULL-handle connect aliases
      // peers to the local heap, and closing that poisons the sticky
      // per-thread HIP error state for later launches
      if (g.peerOpened[p]) (void)hipIpcCloseMemHandle(g.peerBase[p]);
    }
    (void)hipFree(g.heap); (void)hipFree(g.dPeerRecv); (void)hipFree(g.dArrive);
    if (g.hP2pErr) (void)hipHostFree(g.hP2pErr);
  }
  g = State{};
  return FM_OK;
}

int fm_get_compiled_config(int64_t* S, int64_t* H, int64_t* E, int64_t* P,
                           int64_t* PX, int64_t* element_size) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (S) *S = g.S;
  if (H) *H = g.H;
  if (E) *E = g.E;
  if (P) *P = g.P;
  if (PX) *PX = g.PX;
  if (element_size) *element_size = (int64_t)g.esz;
  return FM_OK;
}

int fm_get_num_local_experts(void) { return g.initialized ? g.nLx : -1; }

static int launch_group_gemm(hipStream_t st, int phase, const GemmArgs& a,
                             int M, int nE) {
  const bool hasBias = a.bias != nullptr;
  const int act = (phase == 0) ? a.act : 0;
  const int N = a.N;
  if (g.cfg.dtype == 5 && phase != 3) {
    // MX fp8 grouped GEMM (the gate-logits GEMM stays bf16: gate_w is
    // Element-typed). 256x256 tile when the shape affords it (staging
    // bytes/flop halve: the 128^2 MX tile is glds-rate-bound at ~140
    // GB/s/CU demand vs the ~108 measured ceiling); 128^2 otherwise.
    // Persistent grid amortizes per-job setup.
    GemmArgs aa = a;
    aa.splitK = 1;
    aa.totalJobs = 0;
    // geometry: 128x256 triple-buffered is the measured default
    // wherever N affords 256-wide tiles (cfg3/4/5 shapes: -7..-12% -
    // the 2-stage pipeline's per-tile drain was the stall,
    // profiles/r02_mx_pmc.txt; the 256^2 SE2 tile halves B re-reads but
    // loses more to latency, kept behind FM_MX_GEOM=big for A/B);
    // 128^2 for narrow N
    static const int forceBig = [] {
      const char* e2 = getenv("FM_MX_GEOM");
      return (e2 && e2[0] == 'b') ? 1 : 0;
    }();
    const bool big = forceBig && mxBigGeom(M, N, nE);
    const bool mid = !big && (N >= 256) &&
                     DIVUP(M, 128) * DIVUP(N, 256) * nE >=
                         (g.nCU > 0 ? g.nCU : 256);
    const int bm = big ? 256 : 128, bn = (big || mid) ? 256 : 128;
    static const bool mxS3 = [] {
      const char* e2 = getenv("FM_MX_STAGES");
      return !(e2 && e2[0] == '2');
    }();
    dim3 grid(DIVUP(M, bm), DIVUP(N, bn), nE);
    dim3 block(512);
    {
      const int J = grid.x * grid.y * nE;
      const int resident = (big ? 1 : 2) * (g.nCU > 0 ? g.nCU : 256);
      if (J > resident) {
        aa.totalJobs = J;
        aa.jobsMT = grid.x;
        aa.jobsNT = grid.y;
        grid = dim3(resident, 1, 1);
      }
    }
#define MXGG(PH, AC, HB)                                                      \
    do {                                                                      \
      if (big)                                                                \
        hipLaunchKernelGGL((k_group_gemm_mx<PH, AC, HB, 256, 256>), grid,     \
                           block, 0, st, aa);                                 \
      else if (mid && mxS3)                                                   \
        hipLaunchKernelGGL((k_group_gemm_mx<PH, AC, HB, 128, 256, 3>), grid,  \
                           block, 0, st, aa);                                 \
      else if (mid)                                                           \
        hipLaunchKernelGGL((k_group_gemm_mx<PH, AC, HB, 128, 256>), grid,     \
                           block, 0, st, aa);                                 \
      else                                                                    \
        hipLaunchKernelGGL((k_group_gemm_mx<PH, AC, HB, 128, 128>), grid,     \
                           block, 0, st, aa);                                 \
    } while (0)
    const int sel = phase * 4 + act * 2 + (hasBias ? 1 : 0);
    switch (sel) {
      case 0: MXGG(0, 0, false); break;
      case 1: MXGG(0, 0, true); break;
      case 2: MXGG(0, 1, false); break;
      case 3: MXGG(0, 1, true); break;
      case 4: MXGG(1, 0, false); break;
      case 5: MXGG(1, 0, true); break;
      default:
        if (hasBias) MXGG(2, 0, true);
        else MXGG(2, 0, false);
        break;
    }
#undef MXGG
    FM_HIP_CHECK(hipGetLastError());
    return FM_OK;
  }
  if (g.esz == 2) {
    // tile selection: prefer the deep-pipelined 256-row kernel when the
    // grid still fills the 256 CUs at 1 block/CU. When BN=256 alone
    // cannot fill the chip (small N, e.g. the down GEMM at H=1024), the
    // fp32-atomic combine epilogue makes K-split partials free, so
    // split-K restores occupancy at BN=256's lower staging traffic.
    const int b256 = DIVUP(M, 256) * DIVUP(N, 256) * nE;
    const int b128n = DIVUP(M, 256) * DIVUP(N, 128) * nE;
    const int h256 = DIVUP(M, 128) * DIVUP(N, 256) * nE;  // BM=128 rows
    const int h128 = DIVUP(M, 128) * DIVUP(N, 128) * nE;
    // modes: 0 big(BMxBN 256x256), 1 big(256x128), 3 big(128x256),
    // 4 big(128x128), 2 small synchronous 128x128. Prefer the largest
    // tile whose grid still fills the chip; the BM=128 pipelined tiles
    // cover the many-experts shapes where per-expert M == pEC == 128
    // (configs 3-5 shapes and the EP segments).
    // NOTE: split-K via the atomic combine epilogue is wired up
    // (a.splitK) but DISABLED by default: at config 2 it doubled the
    // 8.3M fp32 atomics and cost more (+13 us) than the BN=256 staging
    // saving bought (profiles/r01). Revisit for huge-K shapes.
    int mode;
    int skf = 1;
    static const int forceMode = [] {
      const char* e = getenv("FM_FORCE_MODE");
      return e ? atoi(e) : -1;
    }();
    static const int noRemap = [] {
      const char* e = getenv("FM_NO_REMAP");
      return (e && e[0] == '1') ? 1 : 0;
    }();
    if (M >= 256 && b256 >= 256) mode = 0;
    // 128x128 runs 2 blocks/CU (65 KB LDS, 71 VGPR) so block-level
    // overlap hides the per-tile sync; prefer it over the BN=128 /
    // BM=128 single-resident tiles whenever the grid can fill 2/CU AND
    // K is not huge (at K=14336 the 1.33x staging of the smaller tile
    // outweighs the overlap - measured on the cfg4 shape). Checked
    // BEFORE mode 1: the cfg2 down GEMM (M=N=1024) runs 69 us in mode 4
    // vs 73 us in mode 1 (measured).
    else if (M >= 128 && h128 >= 512 && a.K <= 8192) mode = 4;
    else if (M >= 256 && b128n >= 256) mode = 1;
    else if (M >= 128 && h256 >= 256) mode = 3;
    else if (M >= 128 && h128 >= 256) mode = 4;
    else mode = 2;
    if (forceMode >= 0) mode = forceMode;
    GemmArgs aa = a;
    aa.splitK = skf;
    aa.noRemap = noRemap;
    aa.totalJobs = 0;
    const int bmSel = (mode == 0 || mode == 1) ? 256 : 128;
    const int bnSel = (mode == 0 || mode == 3) ? 256 : 128;
    dim3 block(mode == 2 ? 256 : 512);
    dim3 grid(DIVUP(M, mode == 2 ? 128 : bmSel),
              DIVUP(N, mode == 2 ? 128 : bnSel), nE * skf);
    // persistent grid for the big-tile modes: launch exactly the number
    // of co-resident blocks (1/CU for the 1-block modes, 2/CU for the
    // 128^2 tile) and let each walk several (m,n,e) tiles - amortizes
    // the per-block setup + prologue staging bubble (measured ~19 us of
    // the 71 us cfg2 up kernel). FM_PERSIST=0 reverts to 1 tile/block.
    static const bool persist = [] {
      const char* e = getenv("FM_PERSIST");
      return !(e && e[0] == '0');
    }();
    if (persist && mode != 2) {
      const int jm = grid.x, jn = grid.y, J = jm * jn * nE * skf;
      const int resident = (mode == 4) ? 512 : 256;
      if (J > resident) {
        aa.totalJobs = J;
        aa.jobsMT = jm;
        aa.jobsNT = jn;
        grid = dim3(resident, 1, 1);
      }
    }
    if (phase == 3) {  // gate logits (fp32 out, B = gate_w)
      // big-E shapes (e.g. cfg5: S=16384, E=256) fill the chip with the
      // pipelined 128x128 tile (zeroing of eC still needs the small
      // kernel's block-0 path, so hand that to a tiny memset instead)
      if (mode != 2 && N >= 128) {
        GemmArgs a3 = a;
        a3.splitK = 1;
        a3.noRemap = noRemap;
        a3.totalJobs = 0;
        if (a3.eC)  // PHASE 3 big kernel does not zero eC; do it here
          FM_HIP_CHECK(hipMemsetAsync(const_cast<uint32_t*>(a3.eC), 0,
                                      (size_t)a3.EC * sizeof(uint32_t), st));
        // K-split to fill the chip when the logits tile grid is narrow
        // (cfg5: 128 x 2 tiles left half the CUs idle): partials
        // atomicAdd into logits32, which is kept zero between forwards
        // by the route's re-zeroing pass
        {
          const int b3 = DIVUP(M, bmSel) * DIVUP(N, bnSel);
          while (a3.splitK < 8 && b3 * a3.splitK * 2 <= 512 &&
                 (a.K / 64) % (a3.splitK * 2) == 0)
            a3.splitK *= 2;
          a3.atomicLogits = (a3.splitK > 1) ? 1 : 0;
        }
        dim3 g3(DIVUP(M, bmSel), DIVUP(N, bnSel), a3.splitK);
        if (g.cfg.dtype == 3) {
          if (mode == 0)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<fp16, 3, 0, false, 256, 256>), g3, dim3(512), 0, st, a3);
          else if (mode == 1)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<fp16, 3, 0, false, 128, 256>), g3, dim3(512), 0, st, a3);
          else if (mode == 3)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<fp16, 3, 0, false, 256, 128>), g3, dim3(512), 0, st, a3);
          else
            hipLaunchKernelGGL((k_group_gemm_bf16_big<fp16, 3, 0, false, 128, 128>), g3, dim3(512), 0, st, a3);
        } else {
          if (mode == 0)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<bf16, 3, 0, false, 256, 256>), g3, dim3(512), 0, st, a3);
          else if (mode == 1)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<bf16, 3, 0, false, 128, 256>), g3, dim3(512), 0, st, a3);
          else if (mode == 3)
            hipLaunchKernelGGL((k_group_gemm_bf16_big<bf16, 3, 0, false, 256, 128>), g3, dim3(512), 0, st, a3);
          else
            hipLaunchKernelGGL((k_group_gemm_bf16_big<bf16, 3, 0, false, 128, 128>), g3, dim3(512), 0, st, a3);
        }
        FM_HIP_CHECK(hipGetLastError());
        return FM_OK;
      }
      // small-E: 128^2 kernel with K-split to fill the chip (the logits
      // grid is only S/128 x E/128 blocks)
      const int b3 = DIVUP(M, 128) * DIVUP(N, 128);
      int sk3 = 1;
      while (sk3 < 8 && b3 * sk3 * 2 <= 512 && (a.K / 64) % (sk3 * 2) == 0)
        sk3 *= 2;
      GemmArgs a3 = a;
      a3.splitK = sk3;
      // logits32 is zero on entry (zeroed at initialize and re-zeroed by
      // k_gate_route after each read)
      dim3 g3(DIVUP(M, 128), DIVUP(N, 128), sk3);
      if (g.cfg.dtype == 3)
        hipLaunchKernelGGL((k_group_gemm_bf16<fp16, 3, 0, false>), g3,
                           dim3(256), 0, st, a3);
      else
        hipLaunchKernelGGL((k_group_gemm_bf16<bf16, 3, 0, false>), g3,
                           dim3(256), 0, st, a3);
      FM_HIP_CHECK(hipGetLastError());
      return FM_OK;
    }
    const int sel = phase * 4 + act * 2 + (hasBias ? 1 : 0);
    // modes 1/3 (the 128-wide tiles) run the triple-buffered staging
    // pipeline by default (LDS 144 KB; counted vmcnt waits, two compute
    // phases of staging latency budget); FM_STAGES=2 reverts for A/B
    // comparison. Modes 0/4 keep double buffering (LDS / occupancy).
    static const bool s3 = [] {
      const char* e = getenv("FM_STAGES");
      return !(e && e[0] == '2');
    }();
#define GG_ET(ET, PH, AC, HB, WET)                                            \
    do {                                                                      \
      if (mode == 0)  /* triple measured -5% here even for fp8 B */          \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 256, 256, WET>),           \
            grid, block, 0, st, aa);                                          \
      else if (mode == 1 && s3)                                               \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 128, 256, WET, 3>),        \
            grid, block, 0, st, aa);                                          \
      else if (mode == 1)                                                     \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 128, 256, WET>),           \
            grid, block, 0, st, aa);                                          \
      else if (mode == 3 && s3)                                               \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 256, 128, WET, 3>),        \
            grid, block, 0, st, aa);                                          \
      else if (mode == 3)                                                     \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 256, 128, WET>),           \
            grid, block, 0, st, aa);                                          \
      else if (mode == 4)                                                     \
        hipLaunchKernelGGL(                                                   \
            (k_group_gemm_bf16_big<ET, PH, AC, HB, 128, 128, WET>),           \
            grid, block, 0, st, aa);                                          \
      else                                                                    \
        hipLaunchKernelGGL((k_group_gemm_bf16<ET, PH, AC, HB, WET>), grid,    \
                           block, 0, st, aa);                                 \
    } while (0)
#define GG_CASE(PH, AC, HB)                                                   \
    do {                                                                      \
      if (g.cfg.dtype == 3) GG_ET(fp16, PH, AC, HB, fp16);                    \
      else if (g.cfg.dtype == 4) GG_ET(bf16, PH, AC, HB, fp8e4m3);            \
      else GG_ET(bf16, PH, AC, HB, bf16);                                     \
    } while (0)
    switch (sel) {
      case 0: GG_CASE(0, 0, false); break;
      case 1: GG_CASE(0, 0, true); break;
      case 2: GG_CASE(0, 1, false); break;
      case 3: GG_CASE(0, 1, true); break;
      case 4: GG_CASE(1, 0, false); break;
      case 5: GG_CASE(1, 0, true); break;
      default:
        if (hasBias) GG_CASE(2, 0, true);
        else GG_CASE(2, 0, false);
        break;
    }
#undef GG_CASE
  } else {
    dim3 block(256);
    dim3 grid(DIVUP(M, 64), DIVUP(N, 64), nE);
    const int sel = phase * 4 + act * 2 + (hasBias ? 1 : 0);
    switch (sel) {
      case 0: hipLaunchKernelGGL((k_group_gemm_f32<0, 0, false>), grid, block, 0, st, a); break;
      case 1: hipLaunchKernelGGL((k_group_gemm_f32<0, 0, true>), grid, block, 0, st, a); break;
      case 2: hipLaunchKernelGGL((k_group_gemm_f32<0, 1, false>), grid, block, 0, st, a); break;
      case 3: hipLaunchKernelGGL((k_group_gemm_f32<0, 1, true>), grid, block, 0, st, a); break;
      case 4: hipLaunchKernelGGL((k_group_gemm_f32<1, 0, false>), grid, block, 0, st, a); break;
      case 5: hipLaunchKernelGGL((k_group_gemm_f32<1, 0, true>), grid, block, 0, st, a); break;
      default:
        if (hasBias) hipLaunchKernelGGL((k_group_gemm_f32<2, 0, true>), grid, block, 0, st, a);
        else hipLaunchKernelGGL((k_group_gemm_f32<2, 0, false>), grid, block, 0, st, a);
        break;
    }
  }
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

static int launch_cast_combine(hipStream_t st, void* moe_out) {
  const int blocks = std::min(g.S, 4096);
#define CC(T, KK)                                                             \
  hipLaunchKernelGGL((k_cast_combine<T, KK>), dim3(blocks), dim3(256), 0, st, \
                     reinterpret_cast<const T*>(g.cbuf), g.kept,              \
                     reinterpret_cast<T*>(moe_out), g.S, g.H)
#define CC_K(T)                                                               \
  switch (g.cfg.expert_top_k) {                                               \
    case 2: CC(T, 2); break;                                                  \
    case 3: CC(T, 3); break;                                                  \
    case 4: CC(T, 4); break;                                                  \
    case 5: CC(T, 5); break;                                                  \
    case 6: CC(T, 6); break;                                                  \
    case 7: CC(T, 7); break;                                                  \
    case 8: CC(T, 8); break;                                                  \
    default: setErr("bad topk"); return FM_ERR_UNSUPPORTED;                   \
  }
  if (g.cfg.dtype == 3) { CC_K(fp16) }
  else if (g.esz == 2) { CC_K(bf16) }
  else { CC_K(float) }
#undef CC_K
#undef CC
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

// dtype 5: quantize bf16 rows into fp8 + per-64 block scales
static int launch_quant_mx(hipStream_t st, const void* in, uint8_t* out,
                           uint8_t* scales, long long nRows, int K) {
  const long long nBlk = nRows * (K / 64);
  const int grid = (int)std::min<long long>(DIVUP(nBlk, 4), 8192);
  hipLaunchKernelGGL(k_quant_mx<bf16>, dim3(grid), dim3(256), 0, st,
                     reinterpret_cast<const bf16*>(in), out, scales, nRows,
                     K);
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

static int moe_forward_impl(hipStream_t st, const void* x, const void* gate_w,
                            const void* expert_w, const void* b_up,
                            const void* b_dn, void* gate_out, void* moe_out,
                            int64_t S, hipEvent_t* evs) {
  // fused single-launch path (the north-star structure); classic
  // multi-kernel path remains for fp32 and as FM_FUSED=0 fallback
  g.lastForwardFused = false;
  if (fusedEnabled() && g.esz == 2 && g.world == 1 && g.cfg.dtype != 5) {
    int rc = moe_forward_fused(st, x, gate_w, expert_w, b_up, b_dn, gate_out,
                               moe_out);
    if (rc != FM_FALLBACK) {
      if (rc == FM_OK) g.lastForwardFused = true;
      return rc;
    }
  }
  // eC is zeroed by the gate logits kernel (block 0)
  // O32 is kept zero across calls (zeroed at initialize; k_cast_out
  // re-zeroes after reading)
  // k==1: zero moe_out so dropped tokens keep zeros (clearState,
  // moe.cuh:30-70); k>1 writes every element via k_cast_out instead
  if (g.cfg.expert_top_k == 1) {
    FM_HIP_CHECK(hipMemsetAsync(moe_out, 0, (size_t)g.S * g.H * g.esz, st));
  }
  if (evs) FM_HIP_CHECK(hipEventRecord(evs[1], st));

  int rc = launch_gate(st, x, gate_w, gate_out, S);
  if (rc != FM_OK) return rc;
  if (evs) FM_HIP_CHECK(hipEventRecord(evs[2], st));

  if (g.cfg.dtype == 5) {
    // MX: quantize the activations once; the gate above read bf16 x
    rc = launch_quant_mx(st, x, g.x8, g.xs, g.S, g.H);
    if (rc != FM_OK) return rc;
  }

  GemmArgs up{};
  up.A = (g.cfg.dtype == 5) ? (const void*)g.x8 : x;
  up.aScales = g.xs;
  up.B = expert_w;
  up.bias = b_up;
  up.out = g.xM;
  up.gate_out = gate_out;
  up.tokenIds = g.tokenIds;
  up.eC = g.eC;
  up.strideAExpert = 0;
  up.strideBExpert = 2LL * g.P * g.H;
  up.strideOExpert = (long long)g.pEC * g.P;
  up.K = g.H; up.N = g.P;
  up.EC = g.EC; up.pEC = g.pEC; up.PX = g.PX;
  up.topk = g.cfg.expert_top_k; up.act = g.cfg.hidden_act;
  up.expertOffset = 0; up.nRows = 0; up.H = g.H;
  const bool mxEpiQuant =
      (g.cfg.dtype == 5) && mxWideN(g.pEC, g.P, g.E);
  if (mxEpiQuant) {
    up.out8 = g.xM8;       // 256-tile up epilogue quantizes in-register
    up.outScales = g.xMs;  // (skips the separate k_quant_mx xM pass)
  }
  rc = launch_group_gemm(st, 0, up, g.pEC, g.E);
  if (rc != FM_OK) return rc;
  if (evs) FM_HIP_CHECK(hipEventRecord(evs[3], st));

  if (g.cfg.dtype == 5 && !mxEpiQuant) {
    // quantize the intermediate activations for the MX down GEMM
    rc = launch_quant_mx(st, g.xM, g.xM8, g.xMs,
                         (long long)g.E * g.pEC, g.P);
    if (rc != FM_OK) return rc;
  }

  GemmArgs dn = up;
  dn.out8 = nullptr;
  dn.outScales = nullptr;
  dn.A = (g.cfg.dtype == 5) ? (const void*)g.xM8 : g.xM;
  dn.aScales = g.xMs;
  dn.B = reinterpret_cast<const char*>(expert_w) + (size_t)g.P * g.H * g.wesz;
  dn.bias = b_dn;
  dn.out = nullptr;
  dn.O32 = reinterpret_cast<float*>(g.cbuf);  // non-atomic combine slots
  dn.moe_out = moe_out;
  dn.strideAExpert = (long long)g.pEC * g.P;
  dn.K = g.P; dn.N = g.H;
  rc = launch_group_gemm(st, 1, dn, g.pEC, g.E);
  if (rc != FM_OK) return rc;
  if (evs) FM_HIP_CHECK(hipEventRecord(evs[4], st));

  if (g.cfg.expert_top_k > 1) {
    int rc2 = launch_cast_combine(st, moe_out);
    if (rc2 != FM_OK) return rc2;
  }
  return FM_OK;
}

int fm_moe_forward(void* stream, const void* x, const void* gate_w,
                   const void* expert_w, const void* b_up, const void* b_dn,
                   void* gate_out, void* moe_out, int64_t S) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (g.world != 1) {
    setErr("fm_moe_forward is the single-rank path; use the staged EP entry points");
    return FM_ERR_STATE;
  }
  if (S != g.S) { setErr("S mismatch vs frozen config"); return FM_ERR_SHAPE; }
  int prc = fusedPoisonCheck();  // a prior fused forward that gave up
  if (prc != FM_OK) return prc;
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);

  // hipGraph fast path: torch's caching allocator hands back the same
  // buffers on steady-state iterations, so after two identical calls we
  // capture the 6-launch sequence once and replay it (saves the
  // per-launch host+boundary overhead; falls back to eager whenever any
  // pointer changes or capture is unavailable).
  void* key[8] = {const_cast<void*>(x), const_cast<void*>(gate_w),
                  const_cast<void*>(expert_w), const_cast<void*>(b_up),
                  const_cast<void*>(b_dn), gate_out, moe_out,
                  reinterpret_cast<void*>(static_cast<uintptr_t>(
                      (uint64_t)S << 1 | (fusedEnabled() ? 1 : 0)))};
  if (g.graphValid && memcmp(key, g.graphKey, sizeof(key)) == 0) {
    if (hipGraphLaunch(g.graphExec, st) == hipSuccess) return FM_OK;
    g.graphValid = false;  // replay failed: rebuild next time
  }
  if (st != nullptr && memcmp(key, g.lastKey, sizeof(key)) == 0) {
    // second consecutive identical call: try to capture
    if (g.graphExec) { (void)hipGraphExecDestroy(g.graphExec); g.graphExec = nullptr; }
    if (hipStreamBeginCapture(st, hipStreamCaptureModeThreadLocal) ==
        hipSuccess) {
      int rc = moe_forward_impl(st, x, gate_w, expert_w, b_up, b_dn, gate_out,
                                moe_out, S, nullptr);
      hipGraph_t graph = nullptr;
      hipError_t ec = hipStreamEndCapture(st, &graph);
      if (rc == FM_OK && ec == hipSuccess && graph) {
        if (hipGraphInstantiate(&g.graphExec, graph, nullptr, nullptr, 0) ==
            hipSuccess) {
          memcpy(g.graphKey, key, sizeof(key));
          g.graphValid = true;
          (void)hipGraphDestroy(graph);
          // the capture did not execute; launch the graph now
          if (hipGraphLaunch(g.graphExec, st) == hipSuccess) return FM_OK;
          g.graphValid = false;
        } else {
          (void)hipGraphDestroy(graph);
        }
      } else if (graph) {
        (void)hipGraphDestroy(graph);
      }
      if (rc != FM_OK) return rc;
      // capture succeeded as a recording but instantiate/launch failed:
      // nothing ran yet - fall through to an eager execution
    }
  }
  memcpy(g.lastKey, key, sizeof(key));
  return moe_forward_impl(st, x, gate_w, expert_w, b_up, b_dn, gate_out,
                          moe_out, S, nullptr);
}

int fm_moe_forward_phased(void* stream, const void* x, const void* gate_w,
                          const void* expert_w, const void* b_up,
                          const void* b_dn, void* gate_out, void* moe_out,
                          int64_t S, float ms[4]) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (g.world != 1) { setErr("single-rank path only"); return FM_ERR_STATE; }
  if (S != g.S) { setErr("S mismatch vs frozen config"); return FM_ERR_SHAPE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  hipEvent_t evs[6];
  for (auto& e : evs) FM_HIP_CHECK(hipEventCreate(&e));
  FM_HIP_CHECK(hipEventRecord(evs[0], st));
  int rc = moe_forward_impl(st, x, gate_w, expert_w, b_up, b_dn, gate_out,
                            moe_out, S, evs);
  if (rc == FM_OK) {
    FM_HIP_CHECK(hipEventRecord(evs[5], st));
    FM_HIP_CHECK(hipStreamSynchronize(st));
    if (g.lastForwardFused) {
      // fused forward: phase times come from the in-kernel wall clocks
      // (one launch has no event boundaries); ms = {gate+route, up GEMM,
      // down GEMM, combine} like the classic path
      int prc = fusedPoisonCheck();
      if (prc != FM_OK) { rc = prc; }
      else {
        FusedCtl hc;
        FM_HIP_CHECK(hipMemcpy(&hc, g.fusedCtl, sizeof(hc),
                               hipMemcpyDeviceToHost));
        const double khz = (double)g.wallClockKHz;
        ms[0] = (float)((double)(hc.clk[1] - hc.clk[0]) / khz);
        ms[1] = (float)((double)(hc.clk[2] - hc.clk[1]) / khz);
        ms[2] = (float)((double)(hc.clk[3] - hc.clk[2]) / khz);
        ms[3] = (float)((double)(hc.clk[4] - hc.clk[3]) / khz);
      }
    } else {
      float pre = 0, gate = 0, upT = 0, dnT = 0, post = 0;
      (void)hipEventElapsedTime(&pre, evs[0], evs[1]);
      (void)hipEventElapsedTime(&gate, evs[1], evs[2]);
      (void)hipEventElapsedTime(&upT, evs[2], evs[3]);
      (void)hipEventElapsedTime(&dnT, evs[3], evs[4]);
      (void)hipEventElapsedTime(&post, evs[4], evs[5]);
      ms[0] = gate; ms[1] = upT; ms[2] = dnT; ms[3] = pre + post;
    }
  }
  for (auto& e : evs) (void)hipEventDestroy(e);
  return rc;
}

int fm_gate_forward(void* stream, const void* x, const void* gate_w,
                    void* gate_out, int64_t S) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (S != g.S) { setErr("S mismatch vs frozen config"); return FM_ERR_SHAPE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  return launch_gate(st, x, gate_w, gate_out, S);  // gate zeroes eC itself
}

int fm_read_routing(void* stream, uint32_t* routed_counts, uint32_t* token_idx,
                    float* prob_sum) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  uint32_t* hEC = (uint32_t*)malloc((size_t)g.E * sizeof(uint32_t));
  TPS* hT = (TPS*)malloc((size_t)g.E * g.pEC * sizeof(TPS));
  if (!hEC || !hT) { free(hEC); free(hT); setErr("oom"); return FM_ERR_HIP; }
  const uint32_t* eCsrc =
      (g.lastForwardFused && g.fusedCtl) ? fusedECPtr() : g.eC;
  hipError_t e1 = hipMemcpyAsync(hEC, eCsrc, (size_t)g.E * sizeof(uint32_t),
                                 hipMemcpyDeviceToHost, st);
  hipError_t e2 = hipMemcpyAsync(hT, g.tokenIds,
                                 (size_t)g.E * g.pEC * sizeof(TPS),
                                 hipMemcpyDeviceToHost, st);
  hipError_t e3 = hipStreamSynchronize(st);
  if (e1 != hipSuccess || e2 != hipSuccess || e3 != hipSuccess) {
    free(hEC); free(hT); setErr("routing D2H failed"); return FM_ERR_HIP;
  }
  for (int e = 0; e < g.E; ++e) {
    const uint32_t r = hEC[e] < (uint32_t)g.EC ? hEC[e] : (uint32_t)g.EC;
    routed_counts[e] = r;
    for (uint32_t i = 0; i < r; ++i) {
      token_idx[(size_t)e * g.EC + i] = tpsTok(hT[(size_t)e * g.pEC + i].tokenIdx);
      prob_sum[(size_t)e * g.EC + i] = hT[(size_t)e * g.pEC + i].probSum;
    }
  }
  free(hEC); free(hT);
  return FM_OK;
}

int fm_expert_ffn(void* stream, const void* rows, const void* expert_w,
                  const void* b_up, const void* b_dn, void* out_rows,
                  int64_t n_rows, int32_t local_e) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (g.cfg.dtype == 5) {
    setErr("dtype 5 (MX fp8) supports fm_moe_forward only this round");
    return FM_ERR_UNSUPPORTED;
  }
  if (n_rows <= 0) return FM_OK;
  if (n_rows > (int64_t)g.nLxAlloc * g.pEC) {
    setErr("n_rows exceeds workspace"); return FM_ERR_SHAPE;
  }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  GemmArgs up{};
  up.A = rows;
  up.B = reinterpret_cast<const char*>(expert_w) +
         (size_t)local_e * 2 * g.P * g.H * g.wesz;
  up.bias = b_up;
  up.out = g.xM;  // scratch [n_rows, P]
  up.tokenIds = nullptr; up.eC = nullptr;
  up.strideAExpert = 0; up.strideBExpert = 0; up.strideOExpert = 0;
  up.K = g.H; up.N = g.P; up.EC = 0; up.pEC = 0; up.PX = g.PX;
  up.topk = 1; up.act = g.cfg.hidden_act; up.expertOffset = 0;
  up.nRows = (int)n_rows; up.H = g.H;
  int rc = launch_group_gemm(st, 0, up, (int)n_rows, 1);
  if (rc != FM_OK) return rc;
  // note: PHASE 0 with tokenIds==nullptr writes out[e=0 stride 0] = xM rows
  GemmArgs dn = up;
  dn.A = g.xM;
  dn.B = reinterpret_cast<const char*>(expert_w) +
         ((size_t)local_e * 2 + 1) * g.P * g.H * g.wesz;
  dn.bias = b_dn;
  dn.out = out_rows;
  dn.K = g.P; dn.N = g.H;
  return launch_group_gemm(st, 2, dn, (int)n_rows, 1);
}

static size_t alignUp(size_t v, size_t a) { return (v + a - 1) / a * a; }

int fm_heap_init(void) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (g.heap) return FM_OK;
  const size_t cells = (size_t)g.E * g.EC * g.H * g.esz;
  g.heapRecvOff = 0;
  g.heapRetOff = alignUp(cells, 256);
  g.heapDFlagOff = alignUp(g.heapRetOff + cells, 256);
  g.heapRFlagOff = alignUp(g.heapDFlagOff + (size_t)g.E * 8, 256);
  g.heapBytes = alignUp(g.heapRFlagOff + (size_t)g.E * 8, 256);
  FM_HIP_CHECK(hipMalloc(&g.heap, g.heapBytes));
  FM_HIP_CHECK(hipMemset(g.heap, 0, g.heapBytes));
  FM_HIP_CHECK(hipMalloc(&g.dPeerRecv, 4 * (size_t)g.world * 8));
  g.dPeerRet = g.dPeerRecv + g.world;
  g.dPeerDFlag = g.dPeerRet + g.world;
  g.dPeerRFlag = g.dPeerDFlag + g.world;
  FM_HIP_CHECK(hipMalloc(&g.dArrive, 2 * (size_t)g.E * 4));
  // zero-copy timeout flag: k_await_flags sets it on a bounded-spin
  // give-up; the host reads it without a stream sync
  FM_HIP_CHECK(hipHostMalloc(&g.hP2pErr, sizeof(uint32_t),
                             hipHostMallocMapped));
  *g.hP2pErr = 0;
  FM_HIP_CHECK(hipHostGetDevicePointer(
      reinterpret_cast<void**>(&g.dP2pErr), g.hP2pErr, 0));
  return FM_OK;
}

int fm_heap_handle(void* out64) {
  if (!g.heap) { setErr("fm_heap_init first"); return FM_ERR_STATE; }
  hipIpcMemHandle_t h;
  FM_HIP_CHECK(hipIpcGetMemHandle(&h, g.heap));
  memcpy(out64, &h, sizeof(h));
  return FM_OK;
}

int fm_heap_connect(const void* handles /* world x 64B, null = local only */) {
  if (!g.heap) { setErr("fm_heap_init first"); return FM_ERR_STATE; }
  uint64_t hRecv[64], hRet[64], hDF[64], hRF[64];
  for (int p = 0; p < g.world; ++p) {
    void* base = nullptr;
    if (p == g.rank || !handles) {
      base = g.heap;
    } else {
      hipIpcMemHandle_t h;
      memcpy(&h, reinterpret_cast<const char*>(handles) + (size_t)p * 64,
             sizeof(h));
      FM_HIP_CHECK(hipIpcOpenMemHandle(&base, h,
                                       hipIpcMemLazyEnablePeerAccess));
      g.peerOpened[p] = true;
    }
    g.peerBase[p] = base;
    hRecv[p] = (uint64_t)(uintptr_t)base + g.heapRecvOff;
    hRet[p] = (uint64_t)(uintptr_t)base + g.heapRetOff;
    hDF[p] = (uint64_t)(uintptr_t)base + g.heapDFlagOff;
    hRF[p] = (uint64_t)(uintptr_t)base + g.heapRFlagOff;
  }
  FM_HIP_CHECK(hipMemcpy(g.dPeerRecv, hRecv, (size_t)g.world * 8,
                         hipMemcpyHostToDevice));
  FM_HIP_CHECK(hipMemcpy(g.dPeerRet, hRet, (size_t)g.world * 8,
                         hipMemcpyHostToDevice));
  FM_HIP_CHECK(hipMemcpy(g.dPeerDFlag, hDF, (size_t)g.world * 8,
                         hipMemcpyHostToDevice));
  FM_HIP_CHECK(hipMemcpy(g.dPeerRFlag, hRF, (size_t)g.world * 8,
                         hipMemcpyHostToDevice));
  return FM_OK;
}

int fm_heap_ptrs(void** recv, void** ret) {
  if (!g.heap) { setErr("fm_heap_init first"); return FM_ERR_STATE; }
  if (recv) *recv = g.heap + g.heapRecvOff;
  if (ret) *ret = g.heap + g.heapRetOff;
  return FM_OK;
}

// bounded-spin budget for the in-kernel flag waits; FM_P2P_SPIN_LOG2
// shrinks it for the forced-timeout test
static long long p2pSpinBound() {
  // re-read per call: the forced-timeout test shrinks it at runtime
  const char* e = getenv("FM_P2P_SPIN_LOG2");
  const int lg = e ? atoi(e) : 26;
  return 1ll << (lg < 4 ? 4 : (lg > 40 ? 40 : lg));
}

// a previously-completed k_await_flags timeout poisons the transport:
// surface it at the next P2P entry (no sync needed - the flag is
// host-mapped) and definitively via fm_p2p_error_check below
static int p2pPoisonCheck() {
  if (g.hP2pErr && *reinterpret_cast<volatile uint32_t*>(g.hP2pErr)) {
    setErr("p2p exchange timed out (k_await_flags gave up; output poisoned)");
    return FM_ERR_HIP;
  }
  return FM_OK;
}

int fm_p2p_error_check(void* stream) {
  if (!g.heap) { setErr("fm_heap_init/connect first"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  FM_HIP_CHECK(hipStreamSynchronize(st));
  const int rc = p2pPoisonCheck();
  if (rc != FM_OK && g.hP2pErr) *g.hP2pErr = 0;  // ack: allow retry
  return rc;
}

int fm_dispatch_p2p(void* stream, const void* x) {
  if (!g.heap) { setErr("fm_heap_init/connect first"); return FM_ERR_STATE; }
  int prc = p2pPoisonCheck();
  if (prc != FM_OK) return prc;
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  g.seq += 1;
  FM_HIP_CHECK(hipMemsetAsync(g.dArrive, 0, 2 * (size_t)g.E * 4, st));
  const int nLx = g.E / g.world;
  dim3 grid(g.EC, g.E);
#define DP2P(T)                                                               \
  hipLaunchKernelGGL(k_dispatch_p2p<T>, grid, dim3(128), 0, st,               \
                     reinterpret_cast<const T*>(x), g.tokenIds, g.eC,         \
                     g.dPeerRecv, g.dPeerDFlag, g.dArrive, g.H, g.EC, g.pEC,  \
                     nLx, g.rank, g.seq)
  if (g.cfg.dtype == 3) DP2P(fp16);
  else if (g.esz == 2) DP2P(bf16);
  else DP2P(float);
#undef DP2P
  FM_HIP_CHECK(hipGetLastError());
  // wait for every source's cells for MY experts
  hipLaunchKernelGGL(
      k_await_flags, dim3(1), dim3(256), 0, st,
      reinterpret_cast<const unsigned long long*>(g.heap + g.heapDFlagOff),
      g.E, g.seq, g.dP2pErr, p2pSpinBound());
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_return_p2p(void* stream, const void* ffn_out) {
  if (!g.heap) { setErr("fm_heap_init/connect first"); return FM_ERR_STATE; }
  int prc = p2pPoisonCheck();
  if (prc != FM_OK) return prc;
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const int nLx = g.E / g.world;
  dim3 grid(g.EC, g.world * nLx);
#define RP2P(T)                                                               \
  hipLaunchKernelGGL(k_return_p2p<T>, grid, dim3(128), 0, st,                 \
                     reinterpret_cast<const T*>(ffn_out), g.dPeerRet,         \
                     g.dPeerRFlag, g.dArrive + g.E, g.H, g.EC, nLx, g.rank,   \
                     g.seq)
  if (g.cfg.dtype == 3) RP2P(fp16);
  else if (g.esz == 2) RP2P(bf16);
  else RP2P(float);
#undef RP2P
  FM_HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(
      k_await_flags, dim3(1), dim3(256), 0, st,
      reinterpret_cast<const unsigned long long*>(g.heap + g.heapRFlagOff),
      g.E, g.seq, g.dP2pErr, p2pSpinBound());
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_pack_dispatch(void* stream, const void* x, void* sendbuf) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 grid(g.EC, g.E);
  if (g.cfg.dtype == 3)
    hipLaunchKernelGGL(k_pack_dispatch<fp16>, grid, dim3(128), 0, st,
                       reinterpret_cast<const fp16*>(x), g.tokenIds, g.eC,
                       reinterpret_cast<fp16*>(sendbuf), g.H, g.EC, g.pEC);
  else if (g.esz == 2)
    hipLaunchKernelGGL(k_pack_dispatch<bf16>, grid, dim3(128), 0, st,
                       reinterpret_cast<const bf16*>(x), g.tokenIds, g.eC,
                       reinterpret_cast<bf16*>(sendbuf), g.H, g.EC, g.pEC);
  else
    hipLaunchKernelGGL(k_pack_dispatch<float>, grid, dim3(128), 0, st,
                       reinterpret_cast<const float*>(x), g.tokenIds, g.eC,
                       reinterpret_cast<float*>(sendbuf), g.H, g.EC, g.pEC);
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_combine_padded(void* stream, const void* returned, const void* gate_out,
                      void* moe_out, int64_t S) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (S != g.S) { setErr("S mismatch"); return FM_ERR_SHAPE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  dim3 grid(g.EC, g.E);
  const int tk = g.cfg.expert_top_k;
  if (tk == 1) {
    // k==1: unscaled direct write path via the slot buffer would be
    // redundant; write slots with scale 1 and reduce with the mask
    // (kept mask is exact, so semantics match the reference's
    // CombineMode::single overwrite of a zeroed output)
  }
#define KCP(T)                                                                \
  hipLaunchKernelGGL(k_combine_padded<T>, grid, dim3(256), 0, st,             \
                     reinterpret_cast<const T*>(returned), g.tokenIds, g.eC,  \
                     reinterpret_cast<const T*>(gate_out),                    \
                     reinterpret_cast<T*>(g.cbuf), g.H, g.EC, g.pEC, g.PX, tk)
  if (g.cfg.dtype == 3) KCP(fp16);
  else if (g.esz == 2) KCP(bf16);
  else KCP(float);
#undef KCP
  FM_HIP_CHECK(hipGetLastError());
  if (tk > 1) return launch_cast_combine(st, moe_out);
  // k==1: each token has exactly one slot; reduce with K=1 semantics
  const size_t n = (size_t)g.S * g.H;
  const int blocks = (int)std::min((size_t)4096, (size_t)DIVUP(n, (size_t)256 * 4));
#define CC1(T)                                                                \
  hipLaunchKernelGGL((k_cast_combine<T, 1>), dim3(blocks), dim3(256), 0, st,  \
                     reinterpret_cast<const T*>(g.cbuf), g.kept,              \
                     reinterpret_cast<T*>(moe_out), g.S, g.H)
  if (g.cfg.dtype == 3) CC1(fp16);
  else if (g.esz == 2) CC1(bf16);
  else CC1(float);
#undef CC1
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_expert_ffn_segments(void* stream, const void* rows,
                           const void* seg_expert_dev, int32_t n_segs,
                           const void* expert_w, void* out_rows) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if ((int64_t)n_segs * g.EC > (int64_t)g.nLxAlloc * g.pEC) {
    setErr("segments exceed xM workspace"); return FM_ERR_SHAPE;
  }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (g.cfg.dtype == 5) {
    // MX: quantize the received rows once; the MX GEMM reads x8/scales
    int qrc = launch_quant_mx(st, rows, g.x8, g.xs,
                              (long long)n_segs * g.EC, g.H);
    if (qrc != FM_OK) return qrc;
  }
  GemmArgs up{};
  up.A = (g.cfg.dtype == 5) ? (const void*)g.x8 : rows;
  up.aScales = g.xs;
  up.B = expert_w;
  up.bias = nullptr;
  up.out = g.xM;
  up.tokenIds = nullptr;
  up.eC = nullptr;
  up.strideAExpert = (long long)g.EC * g.H;
  up.strideBExpert = 2LL * g.P * g.H;
  up.strideOExpert = (long long)g.EC * g.P;
  up.K = g.H;
  up.N = g.P;
  up.EC = 0;
  up.pEC = 0;
  up.PX = g.PX;
  up.topk = 1;
  up.act = g.cfg.hidden_act;
  up.expertOffset = 0;
  up.nRows = g.EC;
  up.H = g.H;
  up.splitK = 1;
  up.segExpert = reinterpret_cast<const int32_t*>(seg_expert_dev);
  const bool mxEpiQuant =
      (g.cfg.dtype == 5) && mxWideN(g.EC, g.P, n_segs);
  if (mxEpiQuant) {
    up.out8 = g.xM8;
    up.outScales = g.xMs;
  }
  int rc = launch_group_gemm(st, 0, up, g.EC, n_segs);
  if (rc != FM_OK) return rc;
  if (g.cfg.dtype == 5 && !mxEpiQuant) {
    int qrc = launch_quant_mx(st, g.xM, g.xM8, g.xMs,
                              (long long)n_segs * g.EC, g.P);
    if (qrc != FM_OK) return qrc;
  }
  GemmArgs dn = up;
  dn.out8 = nullptr;
  dn.outScales = nullptr;
  dn.A = (g.cfg.dtype == 5) ? (const void*)g.xM8 : g.xM;
  dn.aScales = g.xMs;
  dn.B = reinterpret_cast<const char*>(expert_w) + (size_t)g.P * g.H * g.wesz;
  dn.out = out_rows;
  dn.strideAExpert = (long long)g.EC * g.P;
  dn.strideOExpert = (long long)g.EC * g.H;
  dn.K = g.P;
  dn.N = g.H;
  return launch_group_gemm(st, 2, dn, g.EC, n_segs);
}

int fm_expert_ffn_grouped(void* stream, const void* rows,
                          const int64_t* offsets, int32_t n_experts,
                          const void* expert_w, const void* b_up,
                          const void* b_dn, void* out_rows) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  for (int le = 0; le < n_experts; ++le) {
    const int64_t n = offsets[le + 1] - offsets[le];
    if (n <= 0) continue;
    const char* r = reinterpret_cast<const char*>(rows) +
                    (size_t)offsets[le] * g.H * g.esz;
    char* o = reinterpret_cast<char*>(out_rows) +
              (size_t)offsets[le] * g.H * g.esz;
    const void* bu = b_up ? reinterpret_cast<const char*>(b_up) +
                                (size_t)le * g.P * g.esz
                          : nullptr;
    const void* bd = b_dn ? reinterpret_cast<const char*>(b_dn) +
                                (size_t)le * g.H * g.esz
                          : nullptr;
    int rc = fm_expert_ffn(stream, r, expert_w, bu, bd, o, n, le);
    if (rc != FM_OK) return rc;
  }
  return FM_OK;
}

int fm_combine(void* stream, const void* rows, const uint32_t* token_idx,
               const float* scale, int64_t n_rows, int32_t zero_first) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  if (zero_first)
    FM_HIP_CHECK(hipMemsetAsync(g.O32, 0, (size_t)g.S * g.H * sizeof(float), st));
  if (n_rows <= 0) return FM_OK;
  // always accumulate into the fp32 buffer; caller passes scale=1.0 rows
  // for the k==1 unscaled semantics (each token appears at most once)
  if (g.cfg.dtype == 3) {
    hipLaunchKernelGGL((k_combine_rows<fp16, true>), dim3((int)n_rows),
                       dim3(256), 0, st, reinterpret_cast<const fp16*>(rows),
                       token_idx, scale, g.O32, (fp16*)nullptr, g.H);
  } else if (g.esz == 2) {
    hipLaunchKernelGGL((k_combine_rows<bf16, true>), dim3((int)n_rows),
                       dim3(256), 0, st, reinterpret_cast<const bf16*>(rows),
                       token_idx, scale, g.O32, (bf16*)nullptr, g.H);
  } else {
    hipLaunchKernelGGL((k_combine_rows<float, true>), dim3((int)n_rows),
                       dim3(256), 0, st, reinterpret_cast<const float*>(rows),
                       token_idx, scale, g.O32, (float*)nullptr, g.H);
  }
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_combine_finalize(void* stream, void* moe_out, int64_t S) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (S != g.S) { setErr("S mismatch"); return FM_ERR_SHAPE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  const size_t n = (size_t)g.S * g.H;
  const int blocks = (int)min((size_t)2048, DIVUP(n, (size_t)256 * 8));
  if (g.cfg.dtype == 3)
    hipLaunchKernelGGL(k_cast_out<fp16>, dim3(blocks), dim3(256), 0, st, g.O32,
                       reinterpret_cast<fp16*>(moe_out), n);
  else if (g.esz == 2)
    hipLaunchKernelGGL(k_cast_out<bf16>, dim3(blocks), dim3(256), 0, st, g.O32,
                       reinterpret_cast<bf16*>(moe_out), n);
  else
    hipLaunchKernelGGL(k_cast_out<float>, dim3(blocks), dim3(256), 0, st,
                       g.O32, reinterpret_cast<float*>(moe_out), n);
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_export_routing(void* stream, void* routed_dev, void* tps_dev) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(k_export_routing, dim3(g.E), dim3(256), 0, st,
                     g.tokenIds,
                     (g.lastForwardFused && g.fusedCtl) ? fusedECPtr() : g.eC,
                     reinterpret_cast<uint32_t*>(routed_dev),
                     reinterpret_cast<uint32_t*>(tps_dev), g.E, g.EC, g.pEC);
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_read_aux_loss(void* stream, float* gML, float* gMeC) {
  if (!g.initialized) { setErr("not initialized"); return FM_ERR_STATE; }
  if (!g.cfg.is_training) { setErr("aux loss requires is_training=1"); return FM_ERR_STATE; }
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  FM_HIP_CHECK(hipMemcpyAsync(gML, g.gML, (size_t)g.E * sizeof(float),
                              hipMemcpyDeviceToHost, st));
  FM_HIP_CHECK(hipMemcpyAsync(gMeC, g.gMeC, (size_t)g.E * sizeof(float),
                              hipMemcpyDeviceToHost, st));
  FM_HIP_CHECK(hipStreamSynchronize(st));
  return FM_OK;
}

// ---------------------------------------------------------------------------
// Device task-queue primitive (round-2 groundwork; reference semantics:
// subscriber::start polling + scheduler::start matching,
// os/subscriber.cuh:333-451 / os/scheduler.cuh:296-441). A seqlock MPMC
// ring of 128-B task descriptors with generation tags:
//   producer of slot s (gen g = s/ring):  wait seq[pos]==2g  -> write desc
//     -> release-store seq[pos]=2g+1
//   consumer of slot s:                   wait seq[pos]==2g+1 -> read desc
//     -> release-store seq[pos]=2g+2
// Device-scope atomics (single GPU); bounded spins so a logic bug aborts
// instead of hanging the box. The fused persistent kernel
// will drive expert tiles through this queue; here it is validated by a
// standalone probe (fm_debug_taskq): P producer blocks enqueue nTasks
// descriptors, C consumer blocks drain them, each task consumed exactly
// once (checksum + count verified on the host).
// ---------------------------------------------------------------------------

struct __align__(128) FmTask {
  uint32_t kind, payload, slot, gen;
  uint32_t pad[28];
};

__global__ void k_taskq_probe(FmTask* ring, uint32_t* seq, uint32_t ringSz,
                              uint32_t nTasks, uint64_t* tail,
                              uint64_t* claim, uint32_t* sum,
                              uint32_t* consumed, uint32_t* errors) {
  const bool producer = (blockIdx.x & 1) == 0;
  if (threadIdx.x != 0) return;  // block-level protocol; lane 0 drives
  if (producer) {
    while (true) {
      const uint64_t s = atomicAdd(reinterpret_cast<unsigned long long*>(tail),
                                   1ull);
      if (s >= nTasks) break;
      const uint32_t pos = (uint32_t)(s % ringSz);
      const uint32_t gen = (uint32_t)(s / ringSz);
      uint32_t spins = 0;
      while (__hip_atomic_load(seq + pos, __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_AGENT) != 2u * gen) {
        if (++spins > (1u << 26)) { atomicAdd(errors, 1u); return; }
        __builtin_amdgcn_s_sleep(2);
      }
      ring[pos].kind = 1;
      ring[pos].payload = (uint32_t)(s * 2654435761u);  // checksum source
      ring[pos].slot = (uint32_t)s;
      ring[pos].gen = gen;
      __hip_atomic_store(seq + pos, 2u * gen + 1u, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  } else {
    while (true) {
      const uint64_t s = atomicAdd(reinterpret_cast<unsigned long long*>(claim),
                                   1ull);
      if (s >= nTasks) break;
      const uint32_t pos = (uint32_t)(s % ringSz);
      const uint32_t gen = (uint32_t)(s / ringSz);
      uint32_t spins = 0;
      while (__hip_atomic_load(seq + pos, __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_AGENT) != 2u * gen + 1u) {
        if (++spins > (1u << 26)) { atomicAdd(errors, 1u); return; }
        __builtin_amdgcn_s_sleep(2);
      }
      const FmTask t = ring[pos];
      if (t.slot != (uint32_t)s || t.gen != gen) atomicAdd(errors, 1u);
      atomicAdd(sum, t.payload);
      atomicAdd(consumed, 1u);
      __hip_atomic_store(seq + pos, 2u * gen + 2u, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  }
}

// test-only probe (see k_mfma_probe)
int fm_debug_fp8cvt(void* stream, const void* in256, void* out256_f32) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(k_fp8cvt_probe, dim3(1), dim3(32), 0, st,
                     reinterpret_cast<const uint32_t*>(in256),
                     reinterpret_cast<float*>(out256_f32));
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_debug_taskq(void* stream, int ring_sz, long long n_tasks,
                   int blocks, uint32_t* out_sum, uint32_t* out_consumed,
                   uint32_t* out_errors) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  FmTask* ring = nullptr;
  uint32_t* seq = nullptr;
  uint64_t* ctrs = nullptr;  // tail, claim
  uint32_t* acc = nullptr;   // sum, consumed, errors
  FM_HIP_CHECK(hipMalloc(&ring, (size_t)ring_sz * sizeof(FmTask)));
  FM_HIP_CHECK(hipMalloc(&seq, (size_t)ring_sz * sizeof(uint32_t)));
  FM_HIP_CHECK(hipMalloc(&ctrs, 2 * sizeof(uint64_t)));
  FM_HIP_CHECK(hipMalloc(&acc, 3 * sizeof(uint32_t)));
  FM_HIP_CHECK(hipMemsetAsync(seq, 0, (size_t)ring_sz * sizeof(uint32_t), st));
  FM_HIP_CHECK(hipMemsetAsync(ctrs, 0, 2 * sizeof(uint64_t), st));
  FM_HIP_CHECK(hipMemsetAsync(acc, 0, 3 * sizeof(uint32_t), st));
  hipLaunchKernelGGL(k_taskq_probe, dim3(blocks), dim3(64), 0, st, ring, seq,
                     (uint32_t)ring_sz, (uint32_t)n_tasks, ctrs, ctrs + 1,
                     acc, acc + 1, acc + 2);
  FM_HIP_CHECK(hipGetLastError());
  uint32_t h[3];
  FM_HIP_CHECK(hipMemcpyAsync(h, acc, sizeof(h), hipMemcpyDeviceToHost, st));
  FM_HIP_CHECK(hipStreamSynchronize(st));
  if (out_sum) *out_sum = h[0];
  if (out_consumed) *out_consumed = h[1];
  if (out_errors) *out_errors = h[2];
  (void)hipFree(ring); (void)hipFree(seq); (void)hipFree(ctrs); (void)hipFree(acc);
  return FM_OK;
}

int fm_debug_mx_mfma(void* stream, const void* A, const void* B,
                     const void* sa, const void* sb, void* D, int opsel) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
#define MXP(O)                                                                  hipLaunchKernelGGL(k_mx_mfma_probe<O>, dim3(1), dim3(64), 0, st,                                 reinterpret_cast<const uint8_t*>(A),                                          reinterpret_cast<const uint8_t*>(B),                                          reinterpret_cast<const uint8_t*>(sa),                                         reinterpret_cast<const uint8_t*>(sb),                                         reinterpret_cast<float*>(D))
  switch (opsel) {
    case 1: MXP(1); break;
    case 2: MXP(2); break;
    case 3: MXP(3); break;
    default: MXP(0); break;
  }
#undef MXP
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

int fm_debug_mfma(void* stream, const void* A, const void* B, void* D) {
  hipStream_t st = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(k_mfma_probe, dim3(1), dim3(64), 0, st,
                     reinterpret_cast<const bf16*>(A),
                     reinterpret_cast<const bf16*>(B),
                     reinterpret_cast<float*>(D));
  FM_HIP_CHECK(hipGetLastError());
  return FM_OK;
}

}  // extern "C"
