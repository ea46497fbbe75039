// bf16 MFMA GEMM with fused bias+activation epilogue — the inference
// processor's matmul (replaces the reference's python-processor escape hatch
// for ML, reference processor/python.rs:47-98, with a native CDNA4 path).
//
// Structure: 128×128 tile, BK=32, 4 waves (256 threads), each wave owns a
// 64×64 sub-tile = 4×4 fragments of mfma_f32_16x16x32_bf16; staging is
// direct global→LDS via __builtin_amdgcn_global_load_lds width=16 (the
// CDNA4 cp.async analog). B is supplied transposed ([N,K] row-major — the
// torch Linear weight layout), so both A and B^T fragments read LDS with the
// same contiguous-16B pattern (ds_read_b128).
//
// C/D fragment mapping (gfx950, 16x16x32): col = lane&15,
// row = (lane>>4)*4 + reg. Verified against torch.matmul in
// tests/test_gpu_kernels.py with random asymmetric inputs (transpose-detecting).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 32
#define GEMM_THREADS 256

enum Act { ACT_NONE = 0, ACT_RELU = 1, ACT_GELU = 2, ACT_SILU = 3 };

DEV_INLINE float apply_act(float x, int act) {
  switch (act) {
    case ACT_RELU: return fmaxf(x, 0.f);
    case ACT_GELU: {
      // tanh approximation (matches torch.nn.GELU(approximate="tanh")).
      // tanh via the hardware exp (v_exp_f32) — libm tanhf is a software
      // routine that costs ~100 TF on the fc1 epilogue (profiles r18).
      float c = 0.7978845608028654f * (x + 0.044715f * x * x * x);
      c = fminf(fmaxf(c, -10.f), 10.f);  // saturate: no overflow in expf
      float e = __expf(2.f * c);
      return 0.5f * x * (1.f + (e - 1.f) / (e + 1.f));
    }
    case ACT_SILU: return x / (1.f + __expf(-x));
    default: return x;
  }
}

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2)
void gemm_bf16_kernel(const __bf16* __restrict__ A,   // [M,K]
                      const __bf16* __restrict__ Bt,  // [N,K]
                      const float* __restrict__ bias, // [N] or null
                      __bf16* __restrict__ C,         // [M,N]
                      int M, int N, int K, int tiles_n) {
  // XCD-aware bijective block swizzle (8 XCDs; guide §5 m204 variant)
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[BM * BK];  // [128][32] linear
  __shared__ __bf16 Bsm[BN * BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;           // 4 waves: 2×2 over the 128×128 tile
  const int wm = (wid >> 1) * 64;     // wave row offset
  const int wn = (wid & 1) * 64;     // wave col offset

  f32x4 acc[4][4] = {};

  // staging geometry: each global_load_lds call moves 64 lanes × 16 B = 1 KiB
  // per wave; a 8 KiB tile needs 2 calls per wave (iter 0/1).
  // linear LDS byte offset for this lane/iter: wid*1024 + lane*16 + iter*4096
  const int lin0 = wid * 1024 + lane * 16;

  for (int k0 = 0; k0 < K; k0 += BK) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int lin = lin0 + it * 4096;           // byte offset in the 8 KiB tile
      int trow = lin >> 6;                  // tile row (64 B per row)
      int tcol = lin & 63;                  // byte within row
      // A: clamp source row (out-of-range rows produce garbage that only
      // lands in out-of-range C rows, which the epilogue never writes)
      int ga_row = row0 + trow;
      ga_row = ga_row < M ? ga_row : M - 1;
      const char* a_src = (const char*)(A + (int64_t)ga_row * K + k0) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)a_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Asm + (wid * 1024 + it * 4096)),
          16, 0, 0);
      int gb_row = col0 + trow;
      gb_row = gb_row < N ? gb_row : N - 1;
      const char* b_src = (const char*)(Bt + (int64_t)gb_row * K + k0) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)b_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Bsm + (wid * 1024 + it * 4096)),
          16, 0, 0);
    }
    __syncthreads();  // drains vmcnt → LDS tiles ready

    // fragment reads: lane holds 8 contiguous bf16 at
    // [row = sub*16 + (lane&15)][(lane>>4)*8]
    bf16x8 a_frag[4], b_frag[4];
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int m = 0; m < 4; ++m)
      a_frag[m] = *(const bf16x8*)&Asm[(wm + m * 16 + fr) * BK + fk];
#pragma unroll
    for (int n = 0; n < 4; ++n)
      b_frag[n] = *(const bf16x8*)&Bsm[(wn + n * 16 + fr) * BK + fk];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    __syncthreads();
  }

  // epilogue: bias + activation, bf16 store with bounds check
  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

// ---- BK=64 128² tile ---------------------------------------------------------
// Same 128×128 tile as gemm_bf16_kernel but a 64-deep K slice per stage:
// halves the K-loop trip count (and the __syncthreads per K) for the
// K=768-class BERT shapes where the BK=32 kernel leaves ~300 TF on the
// table vs hipBLASLt (profiles r06/r17). LDS 16 KiB/operand (32 KiB total
// → still 2 blocks/CU); staging is 4 global_load_lds iterations per wave;
// the MFMA loop gains an inner ks∈{0,1} step reading fragments at
// [row*64 + ks*32 + fk].
#define BK2 64

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2)
void gemm_bf16_k64_kernel(const __bf16* __restrict__ A,   // [M,K]
                          const __bf16* __restrict__ Bt,  // [N,K]
                          const float* __restrict__ bias, // [N] or null
                          __bf16* __restrict__ C,         // [M,N]
                          int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[BM * BK2];  // [128][64] linear, 16 KiB
  __shared__ __bf16 Bsm[BN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  f32x4 acc[4][4] = {};

  const int lin0 = wid * 1024 + lane * 16;

  for (int k0 = 0; k0 < K; k0 += BK2) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int lin = lin0 + it * 4096;          // byte offset in the 16 KiB tile
      int trow = lin >> 7;                 // 128 B per row (64 bf16)
      int tcol = lin & 127;
      int ga_row = row0 + trow;
      ga_row = ga_row < M ? ga_row : M - 1;
      const char* a_src = (const char*)(A + (int64_t)ga_row * K + k0) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)a_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Asm + lin),
          16, 0, 0);
      int gb_row = col0 + trow;
      gb_row = gb_row < N ? gb_row : N - 1;
      const char* b_src = (const char*)(Bt + (int64_t)gb_row * K + k0) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)b_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Bsm + lin),
          16, 0, 0);
    }
    __syncthreads();

    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m)
        a_frag[m] =
            *(const bf16x8*)&Asm[(wm + m * 16 + fr) * BK2 + ks * 32 + fk];
#pragma unroll
      for (int n = 0; n < 4; ++n)
        b_frag[n] =
            *(const bf16x8*)&Bsm[(wn + n * 16 + fr) * BK2 + ks * 32 + fk];
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

// ---- swizzled BK=64 variant --------------------------------------------------
// The linear [128][64] LDS layout has a 128 B row stride = exactly one bank
// period: the 16 fragment rows a quarter-wave reads per ds_read_b128 alias
// onto the SAME banks (PMC r2: 4.4e5 conflicts/dispatch on the k64 kernel).
// XOR-swizzling the 16 B chunk index by (row & 7) spreads them: chunk' =
// chunk ^ (row & 7). global_load_lds writes a wave's 1 KiB LINEARLY from a
// scalar base, so swizzled staging needs register staging + per-lane
// ds_write_b128 instead (the async direct-to-LDS path cannot express it).
template <int ACT, bool HAS_BIAS, int OCC = 2>
__global__ __launch_bounds__(GEMM_THREADS, OCC)
void gemm_bf16_k64s_kernel(const __bf16* __restrict__ A,   // [M,K]
                           const __bf16* __restrict__ Bt,  // [N,K]
                           const float* __restrict__ bias, // [N] or null
                           __bf16* __restrict__ C,         // [M,N]
                           int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[BM * BK2];
  __shared__ __bf16 Bsm[BN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  f32x4 acc[4][4] = {};

  const int lin0 = wid * 1024 + lane * 16;

  typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;
  for (int k0 = 0; k0 < K; k0 += BK2) {
    // register staging: global 16 B loads → swizzled ds_write per lane
    u32x4 va[4], vb[4];
    int dsts[4];
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int lin = lin0 + it * 4096;
      int trow = lin >> 7;                 // 128 B per row
      int tcol = lin & 127;                // multiple of 16
      dsts[it] = (trow << 7) + (((tcol >> 4) ^ (trow & 7)) << 4);
      int ga_row = row0 + trow;
      ga_row = ga_row < M ? ga_row : M - 1;
      va[it] = *(const u32x4*)((const char*)(A + (int64_t)ga_row * K + k0)
                               + tcol);
      int gb_row = col0 + trow;
      gb_row = gb_row < N ? gb_row : N - 1;
      vb[it] = *(const u32x4*)((const char*)(Bt + (int64_t)gb_row * K + k0)
                               + tcol);
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      *(u32x4*)((char*)Asm + dsts[it]) = va[it];
      *(u32x4*)((char*)Bsm + dsts[it]) = vb[it];
    }
    __syncthreads();

    const int fr = lane & 15;
    const int fk8 = lane >> 4;             // 16 B chunk within the 32-el slice
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        int row = wm + m * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        a_frag[m] = *(const bf16x8*)&Asm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int row = wn + n * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        b_frag[n] = *(const bf16x8*)&Bsm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

// ---- register-prefetch swizzled BK=64 ----------------------------------------
// k64s with the NEXT k-slice's global loads issued right after the LDS-write
// barrier, so their (L2/MALL) latency overlaps the MFMA block. Unlike the
// double-buffered k64d (two LDS buffers: register pressure killed it) this
// keeps ONE LDS buffer and double-buffers only the 16 staging VGPRs.
template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2)
void gemm_bf16_k64p_kernel(const __bf16* __restrict__ A,   // [M,K]
                           const __bf16* __restrict__ Bt,  // [N,K]
                           const float* __restrict__ bias, // [N] or null
                           __bf16* __restrict__ C,         // [M,N]
                           int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[BM * BK2];
  __shared__ __bf16 Bsm[BN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  f32x4 acc[4][4] = {};

  const int lin0 = wid * 1024 + lane * 16;

  typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;
  int dsts[4], arow[4], brow[4], tcols[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int lin = lin0 + it * 4096;
    int trow = lin >> 7;
    int tcol = lin & 127;
    dsts[it] = (trow << 7) + (((tcol >> 4) ^ (trow & 7)) << 4);
    tcols[it] = tcol;
    int ga_row = row0 + trow;
    arow[it] = ga_row < M ? ga_row : M - 1;
    int gb_row = col0 + trow;
    brow[it] = gb_row < N ? gb_row : N - 1;
  }
  u32x4 va[4], vb[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    va[it] = *(const u32x4*)((const char*)(A + (int64_t)arow[it] * K) +
                             tcols[it]);
    vb[it] = *(const u32x4*)((const char*)(Bt + (int64_t)brow[it] * K) +
                             tcols[it]);
  }
  for (int k0 = 0; k0 < K; k0 += BK2) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      *(u32x4*)((char*)Asm + dsts[it]) = va[it];
      *(u32x4*)((char*)Bsm + dsts[it]) = vb[it];
    }
    __syncthreads();
    const int k1 = k0 + BK2;
    if (k1 < K) {  // prefetch next slice; latency hides under the MFMAs
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        va[it] = *(const u32x4*)((const char*)(A + (int64_t)arow[it] * K +
                                               k1) + tcols[it]);
        vb[it] = *(const u32x4*)((const char*)(Bt + (int64_t)brow[it] * K +
                                               k1) + tcols[it]);
      }
    }

    const int fr = lane & 15;
    const int fk8 = lane >> 4;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        int row = wm + m * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        a_frag[m] = *(const bf16x8*)&Asm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int row = wn + n * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        b_frag[n] = *(const bf16x8*)&Bsm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

// ---- wide 256x256 swizzled BK=64 ---------------------------------------------
// Same swizzled-LDS scheme as k64s with a 256x256 tile: 8 waves (each owning
// a 64x128 fragment grid), one block per CU, 64 KB LDS. Quadruple the
// MFLOP-per-byte of the 128x128 tile: A re-read N/256 times and B re-read
// M/256 times, which at short-K fc shapes (fc1: M8192 N3072 K768) halves
// the L2/MALL traffic that bounds the 128x128 kernel.
#define WBM 256
#define WBN 256
template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(512, 1)
void gemm_bf16_k64w_kernel(const __bf16* __restrict__ A,   // [M,K]
                           const __bf16* __restrict__ Bt,  // [N,K]
                           const float* __restrict__ bias, // [N] or null
                           __bf16* __restrict__ C,         // [M,N]
                           int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * WBM, col0 = bn * WBN;

  __shared__ __bf16 Asm[WBM * BK2];
  __shared__ __bf16 Bsm[WBN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;         // 8 waves
  const int wm = (wid >> 1) * 64;   // 4 wave-rows of 64
  const int wn = (wid & 1) * 128;   // 2 wave-cols of 128

  f32x4 acc[4][8] = {};

  const int lin0 = wid * 1024 + lane * 16;

  typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;
  for (int k0 = 0; k0 < K; k0 += BK2) {
    // stage 256x64 of A and of Bt: 32 KB each, 4 chunks per thread per mat
    u32x4 va[4], vb[4];
    int dsts[4];
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int lin = lin0 + it * 8192;
      int trow = lin >> 7;                 // 128 B per LDS row
      int tcol = lin & 127;
      dsts[it] = (trow << 7) + (((tcol >> 4) ^ (trow & 7)) << 4);
      int ga_row = row0 + trow;
      ga_row = ga_row < M ? ga_row : M - 1;
      va[it] = *(const u32x4*)((const char*)(A + (int64_t)ga_row * K + k0)
                               + tcol);
      int gb_row = col0 + trow;
      gb_row = gb_row < N ? gb_row : N - 1;
      vb[it] = *(const u32x4*)((const char*)(Bt + (int64_t)gb_row * K + k0)
                               + tcol);
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      *(u32x4*)((char*)Asm + dsts[it]) = va[it];
      *(u32x4*)((char*)Bsm + dsts[it]) = vb[it];
    }
    __syncthreads();

    const int fr = lane & 15;
    const int fk8 = lane >> 4;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[8];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        int row = wm + m * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        a_frag[m] = *(const bf16x8*)&Asm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int n = 0; n < 8; ++n) {
        int row = wn + n * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        b_frag[n] = *(const bf16x8*)&Bsm[row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 8; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

extern "C" int launch_gemm_bf16_k64w(const void* A, const void* Bt,
                                     const float* bias, void* C, int M,
                                     int N, int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + WBM - 1) / WBM;
  int tiles_n = (N + WBN - 1) / WBN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(512);
#define KWDISPATCH(ACT)                                                      \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64w_kernel<ACT, true><<<grid, block, 0, st>>>(              \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64w_kernel<ACT, false><<<grid, block, 0, st>>>(             \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KWDISPATCH(ACT_RELU); break;
    case ACT_GELU: KWDISPATCH(ACT_GELU); break;
    case ACT_SILU: KWDISPATCH(ACT_SILU); break;
    default: KWDISPATCH(ACT_NONE); break;
  }
#undef KWDISPATCH
  return 0;
}

// ---- double-buffered swizzled BK=64 ------------------------------------------
// Same tile/swizzle as k64s but two LDS buffers: the NEXT K-slice's global
// loads are issued before the current slice's MFMAs so HBM latency hides
// under compute, and the loop needs ONE __syncthreads per slice instead of
// two. LDS 2×32 KiB — still 2 blocks/CU.
template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2)
void gemm_bf16_k64d_kernel(const __bf16* __restrict__ A,   // [M,K]
                           const __bf16* __restrict__ Bt,  // [N,K]
                           const float* __restrict__ bias, // [N] or null
                           __bf16* __restrict__ C,         // [M,N]
                           int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[2][BM * BK2];
  __shared__ __bf16 Bsm[2][BN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  f32x4 acc[4][4] = {};

  const int lin0 = wid * 1024 + lane * 16;
  typedef __attribute__((ext_vector_type(4))) uint32_t u32x4;

  // per-lane staging geometry (constant across slices)
  int trows[4], dsts[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int lin = lin0 + it * 4096;
    int trow = lin >> 7;
    int tcol = lin & 127;
    trows[it] = trow;
    dsts[it] = (trow << 7) + (((tcol >> 4) ^ (trow & 7)) << 4);
  }
  int tcols[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) tcols[it] = (lin0 + it * 4096) & 127;

  auto load_slice = [&](int k0, u32x4* va, u32x4* vb) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int ga_row = row0 + trows[it];
      ga_row = ga_row < M ? ga_row : M - 1;
      va[it] = *(const u32x4*)((const char*)(A + (int64_t)ga_row * K + k0)
                               + tcols[it]);
      int gb_row = col0 + trows[it];
      gb_row = gb_row < N ? gb_row : N - 1;
      vb[it] = *(const u32x4*)((const char*)(Bt + (int64_t)gb_row * K + k0)
                               + tcols[it]);
    }
  };
  auto write_slice = [&](int buf, const u32x4* va, const u32x4* vb) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      *(u32x4*)((char*)Asm[buf] + dsts[it]) = va[it];
      *(u32x4*)((char*)Bsm[buf] + dsts[it]) = vb[it];
    }
  };

  u32x4 va[4], vb[4];
  load_slice(0, va, vb);
  write_slice(0, va, vb);
  __syncthreads();

  const int fr = lane & 15;
  const int fk8 = lane >> 4;
  int buf = 0;
  for (int k0 = 0; k0 < K; k0 += BK2) {
    const bool has_next = k0 + BK2 < K;
    if (has_next) load_slice(k0 + BK2, va, vb);  // hides under the MFMAs
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        int row = wm + m * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        a_frag[m] = *(const bf16x8*)&Asm[buf][row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int row = wn + n * 16 + fr;
        int chunk = (ks * 4 + fk8) ^ (row & 7);
        b_frag[n] = *(const bf16x8*)&Bsm[buf][row * BK2 + chunk * 8];
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    }
    if (has_next) write_slice(buf ^ 1, va, vb);
    buf ^= 1;
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

extern "C" int launch_gemm_bf16_k64d(const void* A, const void* Bt,
                                     const float* bias, void* C, int M,
                                     int N, int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define KDDISPATCH(ACT)                                                      \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64d_kernel<ACT, true><<<grid, block, 0, st>>>(              \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64d_kernel<ACT, false><<<grid, block, 0, st>>>(             \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KDDISPATCH(ACT_RELU); break;
    case ACT_GELU: KDDISPATCH(ACT_GELU); break;
    case ACT_SILU: KDDISPATCH(ACT_SILU); break;
    default: KDDISPATCH(ACT_NONE); break;
  }
#undef KDDISPATCH
  return 0;
}

extern "C" int launch_gemm_bf16_k64s(const void* A, const void* Bt,
                                     const float* bias, void* C, int M,
                                     int N, int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define KSDISPATCH(ACT)                                                      \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64s_kernel<ACT, true><<<grid, block, 0, st>>>(              \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64s_kernel<ACT, false><<<grid, block, 0, st>>>(             \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KSDISPATCH(ACT_RELU); break;
    case ACT_GELU: KSDISPATCH(ACT_GELU); break;
    case ACT_SILU: KSDISPATCH(ACT_SILU); break;
    default: KSDISPATCH(ACT_NONE); break;
  }
#undef KSDISPATCH
  return 0;
}

extern "C" int launch_gemm_bf16_k64p(const void* A, const void* Bt,
                                     const float* bias, void* C, int M,
                                     int N, int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define KPDISPATCH(ACT)                                                      \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64p_kernel<ACT, true><<<grid, block, 0, st>>>(              \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64p_kernel<ACT, false><<<grid, block, 0, st>>>(             \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KPDISPATCH(ACT_RELU); break;
    case ACT_GELU: KPDISPATCH(ACT_GELU); break;
    case ACT_SILU: KPDISPATCH(ACT_SILU); break;
    default: KPDISPATCH(ACT_NONE); break;
  }
#undef KPDISPATCH
  return 0;
}

// occupancy-3 A/B of the same kernel (more inter-block latency overlap if
// the register allocator fits 3 blocks; measured via variant 12)
extern "C" int launch_gemm_bf16_k64s3(const void* A, const void* Bt,
                                      const float* bias, void* C, int M,
                                      int N, int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define KS3DISPATCH(ACT)                                                     \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64s_kernel<ACT, true, 3><<<grid, block, 0, st>>>(           \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64s_kernel<ACT, false, 3><<<grid, block, 0, st>>>(          \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KS3DISPATCH(ACT_RELU); break;
    case ACT_GELU: KS3DISPATCH(ACT_GELU); break;
    case ACT_SILU: KS3DISPATCH(ACT_SILU); break;
    default: KS3DISPATCH(ACT_NONE); break;
  }
#undef KS3DISPATCH
  return 0;
}

extern "C" int launch_gemm_bf16_k64(const void* A, const void* Bt,
                                    const float* bias, void* C, int M, int N,
                                    int K, int act, hipStream_t st) {
  if (K % BK2 != 0) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define KDISPATCH(ACT)                                                       \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_k64_kernel<ACT, true><<<grid, block, 0, st>>>(               \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_k64_kernel<ACT, false><<<grid, block, 0, st>>>(              \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: KDISPATCH(ACT_RELU); break;
    case ACT_GELU: KDISPATCH(ACT_GELU); break;
    case ACT_SILU: KDISPATCH(ACT_SILU); break;
    default: KDISPATCH(ACT_NONE); break;
  }
#undef KDISPATCH
  return 0;
}

// ---- skinny 64×64 tile -------------------------------------------------------
// For narrow outputs (N ≤ a few hundred: the MLP anomaly scorer's 32→256→256
// chain) the 128² tile yields only tiles_m×⌈N/128⌉ workgroups — 128 blocks at
// [8192,256], half the 256 CUs idle. The 64² tile quadruples the grid so the
// chip fills; LDS is 8 KiB → high occupancy hides the short K loop.
// Same wave structure as the 128² kernel scaled down: 4 waves in 2×2, each
// owning a 32×32 quadrant = 2×2 mfma_f32_16x16x32_bf16 fragments.
#define SBM 64
#define SBN 64

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 4)
void gemm_bf16_skinny_kernel(const __bf16* __restrict__ A,   // [M,K]
                             const __bf16* __restrict__ Bt,  // [N,K]
                             const float* __restrict__ bias, // [N] or null
                             __bf16* __restrict__ C,         // [M,N]
                             int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {  // XCD-aware bijective swizzle (guide §5)
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * SBM, col0 = bn * SBN;

  __shared__ __bf16 Asm[SBM * BK];  // [64][32] linear, 4 KiB
  __shared__ __bf16 Bsm[SBN * BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;        // 4 waves: 2×2 over the 64×64 tile
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;

  f32x4 acc[2][2] = {};

  // staging: the 4 KiB operand tile = 4 waves × 1 KiB global_load_lds
  const int lin0 = wid * 1024 + lane * 16;
  const int trow = lin0 >> 6;   // 64 B per row (BK=32 bf16)
  const int tcol = lin0 & 63;

  for (int k0 = 0; k0 < K; k0 += BK) {
    int ga_row = row0 + trow;
    ga_row = ga_row < M ? ga_row : M - 1;
    const char* a_src = (const char*)(A + (int64_t)ga_row * K + k0) + tcol;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)a_src,
        (__attribute__((address_space(3))) uint32_t*)((char*)Asm + lin0),
        16, 0, 0);
    int gb_row = col0 + trow;
    gb_row = gb_row < N ? gb_row : N - 1;
    const char* b_src = (const char*)(Bt + (int64_t)gb_row * K + k0) + tcol;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)b_src,
        (__attribute__((address_space(3))) uint32_t*)((char*)Bsm + lin0),
        16, 0, 0);
    __syncthreads();

    bf16x8 a_frag[2], b_frag[2];
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int m = 0; m < 2; ++m)
      a_frag[m] = *(const bf16x8*)&Asm[(wm + m * 16 + fr) * BK + fk];
#pragma unroll
    for (int n = 0; n < 2; ++n)
      b_frag[n] = *(const bf16x8*)&Bsm[(wn + n * 16 + fr) * BK + fk];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    __syncthreads();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

extern "C" void launch_gemm_bf16_skinny(const void* A, const void* Bt,
                                        const float* bias, void* C, int M,
                                        int N, int K, int act,
                                        hipStream_t st) {
  int tiles_m = (M + SBM - 1) / SBM;
  int tiles_n = (N + SBN - 1) / SBN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define SDISPATCH(ACT)                                                       \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_skinny_kernel<ACT, true><<<grid, block, 0, st>>>(            \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_skinny_kernel<ACT, false><<<grid, block, 0, st>>>(           \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: SDISPATCH(ACT_RELU); break;
    case ACT_GELU: SDISPATCH(ACT_GELU); break;
    case ACT_SILU: SDISPATCH(ACT_SILU); break;
    default: SDISPATCH(ACT_NONE); break;
  }
#undef SDISPATCH
}

// ---- 2-phase double-buffered variant -----------------------------------------
// T3-minimal prefetch (guide §5.5): stage K-tile t+1 into the other LDS
// buffer while computing tile t; ONE vmcnt(0)+barrier per K-tile AFTER the
// MFMAs so the staging loads overlap compute. Pays at short-K / latency-bound
// shapes (BERT: K=768) where block occupancy can't hide the HBM latency.
template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2)
void gemm_bf16_2p_kernel(const __bf16* __restrict__ A,
                         const __bf16* __restrict__ Bt,
                         const float* __restrict__ bias,
                         __bf16* __restrict__ C,
                         int M, int N, int K, int tiles_n) {
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  if (nwg >= 16) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, off = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int bm = bid / tiles_n, bn = bid % tiles_n;
  const int row0 = bm * BM, col0 = bn * BN;

  __shared__ __bf16 Asm[2][BM * BK];
  __shared__ __bf16 Bsm[2][BN * BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;
  const int lin0 = wid * 1024 + lane * 16;
  const int NT = K / BK;

  auto stage = [&](int buf, int t) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int lin = lin0 + it * 4096;
      int trow = lin >> 6;
      int tcol = lin & 63;
      int ga_row = row0 + trow;
      ga_row = ga_row < M ? ga_row : M - 1;
      const char* a_src =
          (const char*)(A + (int64_t)ga_row * K + t * BK) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)a_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Asm[buf] +
                                                        lin),
          16, 0, 0);
      int gb_row = col0 + trow;
      gb_row = gb_row < N ? gb_row : N - 1;
      const char* b_src =
          (const char*)(Bt + (int64_t)gb_row * K + t * BK) + tcol;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)b_src,
          (__attribute__((address_space(3))) uint32_t*)((char*)Bsm[buf] +
                                                        lin),
          16, 0, 0);
    }
  };

  f32x4 acc[4][4] = {};
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_sched_barrier(0);
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);  // prefetch next tile
    bf16x8 a_frag[4], b_frag[4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
      a_frag[m] = *(const bf16x8*)&Asm[cur][(wm + m * 16 + fr) * BK + fk];
#pragma unroll
    for (int n = 0; n < 4; ++n)
      b_frag[n] = *(const bf16x8*)&Bsm[cur][(wn + n * 16 + fr) * BK + fk];
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_sched_barrier(0);
    // next tile must be resident before anyone reads buf cur^1
    asm volatile("s_waitcnt vmcnt(0)");
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int col = col0 + wn + n * 16 + c_col_in_frag;
      if (col >= N) continue;
      float b = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wm + m * 16 + c_row_base + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + b;
        v = apply_act(v, ACT);
        C[(int64_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

template <int ACT>
static void dispatch_2p(const void* A, const void* Bt, const float* bias,
                        void* C, int M, int N, int K, int tiles_n, dim3 grid,
                        dim3 block, hipStream_t st) {
  if (bias)
    gemm_bf16_2p_kernel<ACT, true><<<grid, block, 0, st>>>(
        (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,
        tiles_n);
  else
    gemm_bf16_2p_kernel<ACT, false><<<grid, block, 0, st>>>(
        (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,
        tiles_n);
}

extern "C" {

int launch_gemm_bf16_2p(const void* A, const void* Bt, const float* bias,
                        void* C, int M, int N, int K, int act,
                        hipStream_t st) {
  if (K % BK != 0 || K / BK < 2) return -1;
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n), block(GEMM_THREADS);
  switch (act) {
    case ACT_RELU: dispatch_2p<ACT_RELU>(A, Bt, bias, C, M, N, K, tiles_n,
                                         grid, block, st); break;
    case ACT_GELU: dispatch_2p<ACT_GELU>(A, Bt, bias, C, M, N, K, tiles_n,
                                         grid, block, st); break;
    case ACT_SILU: dispatch_2p<ACT_SILU>(A, Bt, bias, C, M, N, K, tiles_n,
                                         grid, block, st); break;
    default: dispatch_2p<ACT_NONE>(A, Bt, bias, C, M, N, K, tiles_n, grid,
                                   block, st); break;
  }
  return 0;
}

void launch_gemm_bf16(const void* A, const void* Bt, const float* bias,
                      void* C, int M, int N, int K, int act, hipStream_t st) {
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  dim3 block(GEMM_THREADS);
#define DISPATCH(ACT)                                                        \
  do {                                                                       \
    if (bias)                                                                \
      gemm_bf16_kernel<ACT, true><<<grid, block, 0, st>>>(                   \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
    else                                                                     \
      gemm_bf16_kernel<ACT, false><<<grid, block, 0, st>>>(                  \
          (const __bf16*)A, (const __bf16*)Bt, bias, (__bf16*)C, M, N, K,    \
          tiles_n);                                                          \
  } while (0)
  switch (act) {
    case ACT_RELU: DISPATCH(ACT_RELU); break;
    case ACT_GELU: DISPATCH(ACT_GELU); break;
    case ACT_SILU: DISPATCH(ACT_SILU); break;
    default: DISPATCH(ACT_NONE); break;
  }
#undef DISPATCH
}

}  // extern "C"
