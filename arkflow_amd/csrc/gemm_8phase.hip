// bf16 MFMA GEMM, 256×256-tile 8-phase pipelined schedule (gfx950).
//
// Structure (CDNA4 guide §5 "256² 8-phase template"): BM=BN=256, BK=64,
// 8 waves (512 threads) in a 2(M)×4(N) grid, per-wave output 128×64 as
// INTERLEAVED 16×16 fragments (m-frag q at row asel*128 + q*32 + wm*16,
// n-frag r at col bsel*128 + r*64 + wn*16) so each compute phase touches
// exactly one A-half and one B-half of the LDS tile. Per K-tile: 4 quadrant
// phases {ds_read frags ∥ issue one half-tile prefetch → raw s_barrier →
// lgkmcnt → setprio(1) MFMA×16 setprio(0) → raw s_barrier}, with ONE counted
// s_waitcnt vmcnt(4) per K-tile boundary (never 0 in the main loop) — global
// prefetches stay in flight across barriers, 3 half-tiles deep.
//
// Half-slot lifecycle per tile (phase order (A0,B0)(A0,B1)(A1,B1)(A1,B0)):
//   A0 free after P1 → P2 issues A0(t+2); B1 free after P2 → P3 issues
//   B1(t+2); A1/B0 free at the tile boundary → P0/P1 of t+1 issue them.
//   Tile t's last half (B0) is followed by exactly 2 halves (4 loads) in
//   issue order ⇒ vmcnt(4) at the boundary proves tile t resident.
//
// Dispatched for N-tile counts that fill the chip; the 128² kernel
// (gemm_bf16.hip) remains for small shapes and edges.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define P8_THREADS 512
#define P8_BN 256
#define P8_BK 64
// B half-tile = 128 rows × 64 cols bf16 = 16 KiB = 2 global_load_lds/thread;
// A half-tile = (BM/2) rows — 16 KiB at BM=256, 8 KiB at BM=128 (1 load).
#define B_HALF_BYTES (128 * 64 * 2)

enum Act8 { A8_NONE = 0, A8_RELU = 1, A8_GELU = 2, A8_SILU = 3 };

DEV_INLINE float apply_act8(float x, int act) {
  switch (act) {
    case A8_RELU: return fmaxf(x, 0.f);
    case A8_GELU: {
      float c = 0.7978845608028654f * (x + 0.044715f * x * x * x);
      c = fminf(fmaxf(c, -10.f), 10.f);  // saturate: no overflow in expf
      float e = __expf(2.f * c);
      return 0.5f * x * (1.f + (e - 1.f) / (e + 1.f));
    }
    case A8_SILU: return x / (1.f + __expf(-x));
    default: return x;
  }
}

// XOR swizzle (involution, 16B-block preserving): fold the row's low 3 bits
// (byte bits 7-9 at the 128-byte row stride) into bank bits 4-6 — spreads a
// 16-lane column read across 8 bank groups (2-way = free, guide G4/m136).
template <bool SWZ>
DEV_INLINE int swz16x32(int byte_off) {
  return SWZ ? (byte_off ^ (((byte_off >> 7) & 7) << 4)) : byte_off;
}

template <int BMT, int ACT, bool HAS_BIAS, bool SWZ>
__global__ __launch_bounds__(P8_THREADS, 1)
void gemm_bf16_8p_kernel(const __bf16* __restrict__ A,   // [M,K]
                         const __bf16* __restrict__ Bt,  // [N,K]
                         const float* __restrict__ bias,
                         __bf16* __restrict__ C, int M, int N, int K,
                         int tiles_n) {
  constexpr int RH = BMT / 2;            // rows per A half
  constexpr int A_HALF = RH * 64 * 2;    // bytes
  constexpr int A_LOADS = A_HALF / (P8_THREADS * 16);  // insts per thread
  constexpr int QF = BMT / 64;           // m-frags per quadrant phase
  // LDS: 2 buffers × (2 A-halves + 2 B-halves)
  constexpr int BUF_BYTES = 2 * A_HALF + 2 * B_HALF_BYTES;
  extern __shared__ char lds[];
  auto slot = [&](int buf, int is_b, int half) -> char* {
    return lds + (size_t)buf * BUF_BYTES +
           (is_b ? (size_t)(2 * A_HALF + half * B_HALF_BYTES)
                 : (size_t)half * A_HALF);
  };

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {  // bijective XCD swizzle (guide m204)
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BMT, col0 = bn * P8_BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;       // 8 waves
  const int wm = wid >> 2;        // 2 in M
  const int wn = wid & 3;         // 4 in N
  const int fr = lane & 15;
  const int NT = K / P8_BK;       // K-tiles (caller guarantees NT>=3, K%64==0)

  // ---- staging: one half-tile = 2 global_load_lds calls per thread ---------
  // linear LDS byte p = wid*1024 + lane*16 + call*8192; source fetches the
  // data whose swizzled position is p (linear dest + inverse-swz source).
  const int p_base = wid * 1024 + lane * 16;

  auto stage_half = [&](int buf, int is_b, int half, int t) {
    const __bf16* src_base = is_b ? Bt : A;
    int rdim = is_b ? N : M;
    int grow0 = (is_b ? col0 : row0) + half * (is_b ? 128 : RH);
    const int nloads = is_b ? 2 : A_LOADS;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      if (c >= nloads) break;
      int p = p_base + c * 8192;
      int l = swz16x32<SWZ>(p);
      int lrow = l >> 7;          // logical row in the 128×64 half
      int lcolb = l & 127;        // byte within the 128-byte row
      int grow = grow0 + lrow;
      grow = grow < rdim ? grow : rdim - 1;
      const char* src =
          (const char*)(src_base + (int64_t)grow * K + t * P8_BK) + lcolb;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)(slot(buf, is_b, half)
                                                        + p),
          16, 0, 0);
    }
  };

  // issue order within a tile's 4 halves: A0, B1, A1, B0  (see header)
  // stage_seq(t, j): issue the j-th half of tile t
  auto stage_seq = [&](int t, int j) {
    if (t >= NT) return;
    int buf = t & 1;
    switch (j) {
      case 0: stage_half(buf, 0, 0, t); break;  // A0
      case 1: stage_half(buf, 1, 1, t); break;  // B1
      case 2: stage_half(buf, 0, 1, t); break;  // A1
      case 3: stage_half(buf, 1, 0, t); break;  // B0
    }
  };

  f32x4 acc[2 * QF][4] = {};

  // ---- prologue: A0(0) B0(0) B1(0) A1(0) A0(1) B0(1) B1(1) → vmcnt(6) ------
  // (issue order per tile: A0, B0, B1, A1 — j = 0,3,1,2 in stage_seq terms)
  stage_seq(0, 0);  // A0(0)
  stage_seq(0, 3);  // B0(0)
  stage_seq(0, 1);  // B1(0)
  stage_seq(0, 2);  // A1(0)
  stage_seq(1, 0);  // A0(1)
  stage_seq(1, 3);  // B0(1)
  stage_seq(1, 1);  // B1(1)
  if constexpr (BMT == 256) {          // tile 0 fully resident
    asm volatile("s_waitcnt vmcnt(6)");
  } else {
    asm volatile("s_waitcnt vmcnt(5)");
  }
  __builtin_amdgcn_sched_barrier(0);
  __builtin_amdgcn_s_barrier();

  // fragment registers persist across phases: A-half reloaded twice per
  // tile (P0: A0, P2: A1); BOTH B-halves loaded once per tile (P0/P1) and
  // reused at P2/P3 — 20 ds_read_b128 per tile instead of 48.
  bf16x8 a_frag[4][2], b0_frag[2][2], b1_frag[2][2];

  auto load_a = [&](const char* a_half) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      int lrow = q * 32 + wm * 16 + fr;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int off = lrow * 128 + ks * 64 + ((lane >> 4) * 16);
        a_frag[q][ks] = *(const bf16x8*)(a_half + swz16x32<SWZ>(off));
      }
    }
  };
  auto load_b = [&](const char* b_half, bf16x8 (*bf)[2]) {
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      int lrow = r * 64 + wn * 16 + fr;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int off = lrow * 128 + ks * 64 + ((lane >> 4) * 16);
        bf[r][ks] = *(const bf16x8*)(b_half + swz16x32<SWZ>(off));
      }
    }
  };

#define P8_MFMA(ASEL2, BSEL2, BF)                                              do {                                                                            __builtin_amdgcn_s_barrier();                                                 asm volatile("s_waitcnt lgkmcnt(0)");                                         __builtin_amdgcn_sched_barrier(0);                                            __builtin_amdgcn_s_setprio(1);                                                _Pragma("unroll") for (int q = 0; q < 4; ++q)                                     _Pragma("unroll") for (int r = 0; r < 2; ++r)                                     _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                                  acc[(ASEL2)*4 + q][(BSEL2)*2 + r] =                                               __builtin_amdgcn_mfma_f32_16x16x32_bf16(                                          a_frag[q][ks], (BF)[r][ks],                                                   acc[(ASEL2)*4 + q][(BSEL2)*2 + r], 0, 0, 0);              __builtin_amdgcn_s_setprio(0);                                                __builtin_amdgcn_sched_barrier(0);                                            __builtin_amdgcn_s_barrier();                                               } while (0)

  for (int t = 0; t < NT; ++t) {
    const int buf = t & 1;
    const char* a0_half = slot(buf, 0, 0);
    const char* a1_half = slot(buf, 0, 1);
    const char* b0_half = slot(buf, 1, 0);
    const char* b1_half = slot(buf, 1, 1);

    // P0 (A0,B0): load a(A0) + b0 + b1; prefetch A1(t+1)
    load_a(a0_half);
    load_b(b0_half, b0_frag);
    stage_seq(t + 1, 2);
    P8_MFMA(0, 0, b0_frag);

    // P1 (A0,B1): load b1; prefetch A0(t+2) (A0 slot LDS-read done at P0)
    load_b(b1_half, b1_frag);
    stage_seq(t + 2, 0);
    P8_MFMA(0, 1, b1_frag);

    // P2 (A1,B1): load a(A1); prefetch B0(t+2)
    load_a(a1_half);
    stage_seq(t + 2, 3);
    P8_MFMA(1, 1, b1_frag);

    // P3 (A1,B0): all frags in registers; prefetch B1(t+2)
    stage_seq(t + 2, 1);
    P8_MFMA(1, 0, b0_frag);

    // K-tile boundary: tile t+1's last half (A1) issued at P0 above is
    // followed by exactly 3 halves (A0+B0+B1 of t+2) ⇒ counted wait.
    if (t + 2 < NT) {
      if constexpr (BMT == 256) {
        asm volatile("s_waitcnt vmcnt(6)");
      } else {
        asm volatile("s_waitcnt vmcnt(5)");
      }
    } else {
      asm volatile("s_waitcnt vmcnt(0)");
    }
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_barrier();
  }
#undef P8_MFMA

  // ---- epilogue: bias + activation, bf16 stores (bounds-checked) -----------
  const int c_row_in_frag = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2 * QF; ++i) {
    int asel = i / QF, q = i % QF;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int bsel = j >> 1, r2 = j & 1;
      int col = col0 + bsel * 128 + r2 * 64 + wn * 16 + fr;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int row = row0 + asel * RH + q * 32 + wm * 16 + c_row_in_frag + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr] + b;
        C[(int64_t)row * N + col] = (__bf16)apply_act8(v, ACT);
      }
    }
  }
}

template <int BMT, int ACT, bool HAS_BIAS, bool SWZ>
static void launch_one_8p(const void* A, const void* Bt, const float* bias,
                          void* C, int M, int N, int K, int tiles_n,
                          dim3 grid, dim3 block, size_t lds_bytes,
                          hipStream_t st) {
  static bool attr_done = false;
  if (!attr_done) {
    hipFuncSetAttribute(
        (const void*)gemm_bf16_8p_kernel<BMT, ACT, HAS_BIAS, SWZ>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);
    attr_done = true;
  }
  gemm_bf16_8p_kernel<BMT, ACT, HAS_BIAS, SWZ><<<grid, block, lds_bytes,
                                                 st>>>(
      (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,
      tiles_n);
}

template <int BMT, int ACT>
static void dispatch_bias_swz(const void* A, const void* Bt,
                              const float* bias, void* C, int M, int N,
                              int K, int tiles_n, dim3 grid, dim3 block,
                              size_t lds, int swz, hipStream_t st) {
  if (bias) {
    if (swz)
      launch_one_8p<BMT, ACT, true, true>(A, Bt, bias, C, M, N, K, tiles_n,
                                          grid, block, lds, st);
    else
      launch_one_8p<BMT, ACT, true, false>(A, Bt, bias, C, M, N, K, tiles_n,
                                           grid, block, lds, st);
  } else {
    if (swz)
      launch_one_8p<BMT, ACT, false, true>(A, Bt, bias, C, M, N, K, tiles_n,
                                           grid, block, lds, st);
    else
      launch_one_8p<BMT, ACT, false, false>(A, Bt, bias, C, M, N, K,
                                            tiles_n, grid, block, lds, st);
  }
}

template <int BMT>
static void dispatch_act(const void* A, const void* Bt, const float* bias,
                         void* C, int M, int N, int K, int act, int swz,
                         hipStream_t st) {
  int tiles_m = (M + BMT - 1) / BMT;
  int tiles_n = (N + P8_BN - 1) / P8_BN;
  size_t lds = 2 * (2 * (size_t)(BMT / 2) * 64 * 2 + 2 * B_HALF_BYTES);
  dim3 grid(tiles_m * tiles_n), block(P8_THREADS);
  switch (act) {
    case A8_RELU:
      dispatch_bias_swz<BMT, A8_RELU>(A, Bt, bias, C, M, N, K, tiles_n, grid,
                                      block, lds, swz, st);
      break;
    case A8_GELU:
      dispatch_bias_swz<BMT, A8_GELU>(A, Bt, bias, C, M, N, K, tiles_n, grid,
                                      block, lds, swz, st);
      break;
    case A8_SILU:
      dispatch_bias_swz<BMT, A8_SILU>(A, Bt, bias, C, M, N, K, tiles_n, grid,
                                      block, lds, swz, st);
      break;
    default:
      dispatch_bias_swz<BMT, A8_NONE>(A, Bt, bias, C, M, N, K, tiles_n, grid,
                                      block, lds, swz, st);
      break;
  }
}

extern "C" {

// returns 0 if dispatched, -1 if the shape doesn't fit this kernel
int launch_gemm_bf16_8p(const void* A, const void* Bt, const float* bias,
                        void* C, int M, int N, int K, int act, int swz,
                        hipStream_t st) {
  if (K % P8_BK != 0 || K / P8_BK < 3) return -1;
  // swz>=2 forces the BM=128 tile (chip-filling at narrow N)
  if (swz >= 2) {
    dispatch_act<128>(A, Bt, bias, C, M, N, K, act, swz == 3 ? 1 : 0, st);
  } else {
    dispatch_act<256>(A, Bt, bias, C, M, N, K, act, swz, st);
  }
  return 0;
}

}  // extern "C"
