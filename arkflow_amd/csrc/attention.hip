// Fused bf16 attention for the BERT inference processor: softmax(Q·K^T)·V
// in ONE kernel, one workgroup per (batch, head). Sized for encoder shapes
// (S=128, D=64: BERT-base). MFMA 16x16x32 for both QK^T and P·V; the softmax
// runs entirely in registers (row max/sum via quarter-wave shfl_xor over the
// MFMA C-fragment layout: col=lane&15, row=(lane>>4)*4+reg).
//
// LDS: K [S][D+4] + V^T [D][S+4] staged once (padded rows: +4 bf16 = 2-bank
// row stride → conflict-free ds_read_b128), P bf16 [32][S+4] per wave.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define ATTN_THREADS 256

// Strided layout: row s of head h in batch b lives at
// base + ((int64_t)b*S + s)*ld + h*D. With Hd=1/ld=D this is the packed
// [BH, S, D] layout; with Hd=H/ld=3*H*D the kernel reads q/k/v DIRECTLY out
// of the [B,S,3,H,D] QKV-projection tensor (no transpose copies), and with
// ldo=H*D writes O back in [B,S,H*D] so the next linear consumes it as-is.
template <int S, int D, int PAD = 4>
__global__ __launch_bounds__(ATTN_THREADS, 1)
void attention_kernel(const __bf16* __restrict__ Q,
                      const __bf16* __restrict__ K,
                      const __bf16* __restrict__ V,
                      __bf16* __restrict__ O, float scale,
                      int ldq, int ldo, int Hd) {
  constexpr int DP = D + PAD;  // padded row strides (bank spread)
  constexpr int SP = S + PAD;
  constexpr int QROWS = 32;   // q rows per wave
  constexpr int NWAVE = ATTN_THREADS / WAVE;
  static_assert(S == NWAVE * QROWS, "one block covers all S rows");

  extern __shared__ char smem[];
  __bf16* K_lds = (__bf16*)smem;                       // [S][DP]
  __bf16* Vt_lds = K_lds + S * DP;                     // [D][SP]
  __bf16* P_lds = Vt_lds + D * SP;                     // [NWAVE][QROWS][SP]

  const int bh = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int b = bh / Hd, h = bh % Hd;
  const __bf16* Qb = Q + (int64_t)b * S * ldq + h * D;
  const __bf16* Kb = K + (int64_t)b * S * ldq + h * D;
  const __bf16* Vb = V + (int64_t)b * S * ldq + h * D;
  __bf16* Ob = O + (int64_t)b * S * ldo + h * D;

  // ---- stage K and V^T (coalesced global reads) -----------------------------
  for (int i = tid; i < S * D / 2; i += ATTN_THREADS) {
    // 2 elements per thread via 32-bit loads
    int s = (i * 2) / D, d = (i * 2) % D;
    uint32_t kv = *(const uint32_t*)(Kb + (int64_t)s * ldq + d);
    *(uint32_t*)(&K_lds[s * DP + d]) = kv;
    uint32_t vv = *(const uint32_t*)(Vb + (int64_t)s * ldq + d);
    __bf16 v0 = ((const __bf16*)&vv)[0], v1 = ((const __bf16*)&vv)[1];
    Vt_lds[d * SP + s] = v0;
    Vt_lds[(d + 1) * SP + s] = v1;
  }

  // ---- Q fragments (held in registers for the whole kernel) ----------------
  const int q0 = wid * QROWS;
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;
  bf16x8 q_frag[2][D / 32];
#pragma unroll
  for (int am = 0; am < 2; ++am)
#pragma unroll
    for (int ks = 0; ks < D / 32; ++ks)
      q_frag[am][ks] = *(const bf16x8*)(
          Qb + (int64_t)(q0 + am * 16 + fr) * ldq + ks * 32 + fk);

  __syncthreads();  // K/Vt staged

  // ---- scores = Q·K^T : acc[am][nf] covers rows 32 × cols S -----------------
  constexpr int NF = S / 16;
  f32x4 acc[2][NF] = {};
#pragma unroll
  for (int nf = 0; nf < NF; ++nf) {
#pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      bf16x8 b_frag =
          *(const bf16x8*)(&K_lds[(nf * 16 + fr) * DP + ks * 32 + fk]);
#pragma unroll
      for (int am = 0; am < 2; ++am)
        acc[am][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[am][ks], b_frag, acc[am][nf], 0, 0, 0);
    }
  }

  // ---- in-register softmax over each score row ------------------------------
  // lane holds, for fixed (am, r), one value per nf → row = quarter-wave's
  // 16 lanes × NF regs. reduce: per-lane over nf, then shfl_xor {1,2,4,8}.
  __bf16* Pw = P_lds + wid * QROWS * SP;
#pragma unroll
  for (int am = 0; am < 2; ++am) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float rmax = -INFINITY;
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        rmax = fmaxf(rmax, acc[am][nf][r] * scale);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rmax = fmaxf(rmax, __shfl_xor(rmax, off, 64));
      float rsum = 0.f;
      float p[NF];
#pragma unroll
      for (int nf = 0; nf < NF; ++nf) {
        p[nf] = __expf(acc[am][nf][r] * scale - rmax);
        rsum += p[nf];
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rsum += __shfl_xor(rsum, off, 64);
      float inv = 1.f / rsum;
      int prow = am * 16 + (lane >> 4) * 4 + r;
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        Pw[prow * SP + nf * 16 + fr] = (__bf16)(p[nf] * inv);
    }
  }
  // P is consumed only by this wave; compiler orders the LDS ops.

  // ---- out = P·V ------------------------------------------------------------
  constexpr int ND = D / 16;
  f32x4 acc2[2][ND] = {};
#pragma unroll
  for (int ks = 0; ks < S / 32; ++ks) {
    bf16x8 a_frag[2];
#pragma unroll
    for (int am = 0; am < 2; ++am)
      a_frag[am] = *(const bf16x8*)(&Pw[(am * 16 + fr) * SP + ks * 32 + fk]);
#pragma unroll
    for (int nd = 0; nd < ND; ++nd) {
      bf16x8 b_frag =
          *(const bf16x8*)(&Vt_lds[(nd * 16 + fr) * SP + ks * 32 + fk]);
#pragma unroll
      for (int am = 0; am < 2; ++am)
        acc2[am][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[am], b_frag, acc2[am][nd], 0, 0, 0);
    }
  }

  // ---- write O --------------------------------------------------------------
#pragma unroll
  for (int am = 0; am < 2; ++am)
#pragma unroll
    for (int nd = 0; nd < ND; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = q0 + am * 16 + (lane >> 4) * 4 + r;
        int col = nd * 16 + fr;
        Ob[(int64_t)row * ldo + col] = (__bf16)acc2[am][nd][r];
      }
}

extern "C" int launch_attention_flash(const void*, const void*,
                                      const void*, void*, int, int, int,
                                      float, int, int, int, hipStream_t);

extern "C" {

// returns 0 on success, -1 if the (S, D) shape is unsupported
int launch_attention_bf16(const void* Q, const void* K, const void* V,
                          void* O, int BH, int S, int D, float scale,
                          hipStream_t st) {
  if (S == 128 && D == 64) {
    // PAD=8: the +4 row pad measured ~590K LDS bank conflicts/dispatch;
    // +8 runs 28.8→19.8 µs at BERT shape (pad sweep, profiles r2)
    constexpr int SS = 128, DD = 64, PD = 8;
    size_t lds = (SS * (DD + PD) + DD * (SS + PD) + 4 * 32 * (SS + PD)) *
                 sizeof(__bf16);
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute(
          (const void*)attention_kernel<SS, DD, PD>,
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
      attr_set = true;
    }
    attention_kernel<SS, DD, PD><<<BH, ATTN_THREADS, lds, st>>>(
        (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V, (__bf16*)O,
        scale, DD, DD, 1);
    return 0;
  }
  // longer sequences (S % 64 == 0, D=64): flash-tiled kernel
  return launch_attention_flash(Q, K, V, O, BH, S, D, scale, D, D, 1, st);
}

// qkv: [B, S, 3, H, D] contiguous (the QKV linear's natural output);
// O: [B, S, H*D]. No transpose copies on either side.
// pad: LDS row-padding in bf16 elements (pad sweep — NOTES r2: the +4
// layout still measured ~590K bank conflicts/dispatch).
}  // extern "C" (template below needs C++ linkage)

// ---- flash-tiled attention: S > 128 (KV streamed in 64-col tiles) ----------
// One workgroup per (batch·head, 128-row Q block); 4 waves × 32 q rows each
// (same fragment layout as the full-S kernel). KV tiles of 64 columns are
// staged in LDS; softmax is the running-rescale form (m/l state per row in
// registers): o ← o·e^{m−m'} + P·Vᵗ, l ← l·e^{m−m'} + Σp. Covers any
// S % 64 == 0 at D=64; the S=128 full-S kernel stays the BERT fast path.
#define FA_QROWS 32
#define FA_KT 64   // kv tile columns
#define FA_PAD 8

template <int D>
__global__ __launch_bounds__(ATTN_THREADS, 2)
void attention_flash_kernel(const __bf16* __restrict__ Q,
                            const __bf16* __restrict__ K,
                            const __bf16* __restrict__ V,
                            __bf16* __restrict__ O, float scale, int S,
                            int ldq, int ldo, int Hd, int qblocks) {
  constexpr int DP = D + FA_PAD;
  constexpr int KTP = FA_KT + FA_PAD;
  __shared__ __bf16 K_lds[FA_KT * DP];        // kv tile, row-major [64][DP]
  __shared__ __bf16 Vt_lds[D * KTP];          // tile of Vᵗ [D][64+pad]
  __shared__ __bf16 P_lds[4][FA_QROWS * KTP]; // per-wave P [32][64+pad]

  const int bh = blockIdx.x / qblocks;
  const int qb = blockIdx.x % qblocks;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int b = bh / Hd, h = bh % Hd;
  const __bf16* Qb = Q + (int64_t)b * S * ldq + h * D;
  const __bf16* Kb = K + (int64_t)b * S * ldq + h * D;
  const __bf16* Vb = V + (int64_t)b * S * ldq + h * D;
  __bf16* Ob = O + (int64_t)b * S * ldo + h * D;

  const int q0 = qb * 128 + wid * FA_QROWS;   // this wave's first q row
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  // Q fragments in registers for the wave's 32 rows (rows ≥ S clamp to S-1;
  // their outputs are never stored)
  bf16x8 q_frag[2][D / 32];
#pragma unroll
  for (int am = 0; am < 2; ++am)
#pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      int row = q0 + am * 16 + fr;
      row = row < S ? row : S - 1;
      q_frag[am][ks] =
          *(const bf16x8*)(Qb + (int64_t)row * ldq + ks * 32 + fk);
    }

  // running state: per (am, r) row of this lane's quarter
  float m_run[2][4], l_run[2][4];
  f32x4 o_acc[2][D / 16] = {};
#pragma unroll
  for (int am = 0; am < 2; ++am)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[am][r] = -INFINITY;
      l_run[am][r] = 0.f;
    }

  for (int kt = 0; kt < S; kt += FA_KT) {
    // stage K tile + Vᵗ tile (cooperative, 2 bf16 per thread per step)
    for (int i = tid; i < FA_KT * D / 2; i += ATTN_THREADS) {
      int s = (i * 2) / D, d = (i * 2) % D;
      uint32_t kv = *(const uint32_t*)(Kb + (int64_t)(kt + s) * ldq + d);
      *(uint32_t*)(&K_lds[s * DP + d]) = kv;
      uint32_t vv = *(const uint32_t*)(Vb + (int64_t)(kt + s) * ldq + d);
      __bf16 v0 = ((const __bf16*)&vv)[0], v1 = ((const __bf16*)&vv)[1];
      Vt_lds[d * KTP + s] = v0;
      Vt_lds[(d + 1) * KTP + s] = v1;
    }
    __syncthreads();

    // scores tile: [32 q rows][64 kv cols] per wave = acc[2][4] frags
    f32x4 acc[2][FA_KT / 16] = {};
#pragma unroll
    for (int nf = 0; nf < FA_KT / 16; ++nf) {
#pragma unroll
      for (int ks = 0; ks < D / 32; ++ks) {
        bf16x8 b_frag =
            *(const bf16x8*)(&K_lds[(nf * 16 + fr) * DP + ks * 32 + fk]);
#pragma unroll
        for (int am = 0; am < 2; ++am)
          acc[am][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[am][ks], b_frag, acc[am][nf], 0, 0, 0);
      }
    }

    // running softmax + stage P
    __bf16* Pw = P_lds[wid];
#pragma unroll
    for (int am = 0; am < 2; ++am) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float tmax = -INFINITY;
#pragma unroll
        for (int nf = 0; nf < FA_KT / 16; ++nf)
          tmax = fmaxf(tmax, acc[am][nf][r] * scale);
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          tmax = fmaxf(tmax, __shfl_xor(tmax, off, 64));
        float m_new = fmaxf(m_run[am][r], tmax);
        float rescale = __expf(m_run[am][r] - m_new);
        float psum = 0.f;
        float p[FA_KT / 16];
#pragma unroll
        for (int nf = 0; nf < FA_KT / 16; ++nf) {
          p[nf] = __expf(acc[am][nf][r] * scale - m_new);
          psum += p[nf];
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          psum += __shfl_xor(psum, off, 64);
        l_run[am][r] = l_run[am][r] * rescale + psum;
        m_run[am][r] = m_new;
        // rescale accumulated output for this row
#pragma unroll
        for (int nd = 0; nd < D / 16; ++nd) o_acc[am][nd][r] *= rescale;
        int prow = am * 16 + (lane >> 4) * 4 + r;
#pragma unroll
        for (int nf = 0; nf < FA_KT / 16; ++nf)
          Pw[prow * KTP + nf * 16 + fr] = (__bf16)p[nf];
      }
    }
    // o += P·Vᵗ over this tile
#pragma unroll
    for (int ks = 0; ks < FA_KT / 32; ++ks) {
      bf16x8 a_frag[2];
#pragma unroll
      for (int am = 0; am < 2; ++am)
        a_frag[am] =
            *(const bf16x8*)(&Pw[(am * 16 + fr) * KTP + ks * 32 + fk]);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        bf16x8 b_frag =
            *(const bf16x8*)(&Vt_lds[(nd * 16 + fr) * KTP + ks * 32 + fk]);
#pragma unroll
        for (int am = 0; am < 2; ++am)
          o_acc[am][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[am], b_frag, o_acc[am][nd], 0, 0, 0);
      }
    }
    __syncthreads();  // K/Vt tiles reused next iteration
  }

  // epilogue: O = o / l
  const int c_col = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int am = 0; am < 2; ++am)
#pragma unroll
    for (int nd = 0; nd < D / 16; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = q0 + am * 16 + c_row_base + r;
        if (row >= S) continue;
        float inv = l_run[am][r] > 0.f ? 1.f / l_run[am][r] : 0.f;
        Ob[(int64_t)row * ldo + nd * 16 + c_col] =
            (__bf16)(o_acc[am][nd][r] * inv);
      }
}

template <int PAD>
static int qkv_launch_pad(const void* QKV, void* O, int B, int H, int S,
                          int D, float scale, hipStream_t st) {
  constexpr int SS = 128, DD = 64;
  size_t lds = (SS * (DD + PAD) + DD * (SS + PAD) + 4 * 32 * (SS + PAD)) *
               sizeof(__bf16);
  static bool attr_set2 = false;
  if (!attr_set2) {
    hipFuncSetAttribute(
        (const void*)attention_kernel<SS, DD, PAD>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    attr_set2 = true;
  }
  const __bf16* base = (const __bf16*)QKV;
  int ld = 3 * H * D;
  attention_kernel<SS, DD, PAD><<<B * H, ATTN_THREADS, lds, st>>>(
      base, base + (int64_t)H * D, base + (int64_t)2 * H * D, (__bf16*)O,
      scale, ld, H * D, H);
  return 0;
}

extern "C" int launch_attention_flash(const void* Q, const void* K,
                                      const void* V, void* O, int BH, int S,
                                      int D, float scale, int ldq, int ldo,
                                      int Hd, hipStream_t st) {
  if (D != 64 || S % FA_KT != 0) return -1;
  int qblocks = (S + 127) / 128;
  attention_flash_kernel<64><<<BH * qblocks, ATTN_THREADS, 0, st>>>(
      (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V, (__bf16*)O,
      scale, S, ldq, ldo, Hd, qblocks);
  return 0;
}

extern "C" int launch_attention_qkv_bf16_pad(const void* QKV, void* O,
                                             int B, int H, int S, int D,
                                             float scale, int pad,
                                             hipStream_t st) {
  if (!(S == 128 && D == 64)) {
    // longer sequences: flash-tiled kernel, same strided-QKV layout
    const __bf16* base = (const __bf16*)QKV;
    return launch_attention_flash(base, base + (int64_t)H * D,
                                  base + (int64_t)2 * H * D, O, B * H, S, D,
                                  scale, 3 * H * D, H * D, H, st);
  }
  switch (pad) {
    case 4: return qkv_launch_pad<4>(QKV, O, B, H, S, D, scale, st);
    case 16: return qkv_launch_pad<16>(QKV, O, B, H, S, D, scale, st);
    case 20: return qkv_launch_pad<20>(QKV, O, B, H, S, D, scale, st);
    default: return qkv_launch_pad<8>(QKV, O, B, H, S, D, scale, st);
  }
}

extern "C" int launch_attention_qkv_bf16(const void* QKV, void* O, int B,
                                         int H, int S, int D, float scale,
                                         hipStream_t st) {
  return launch_attention_qkv_bf16_pad(QKV, O, B, H, S, D, scale, 8, st);
}
