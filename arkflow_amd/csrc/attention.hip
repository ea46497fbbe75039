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
  return -1;
}

// qkv: [B, S, 3, H, D] contiguous (the QKV linear's natural output);
// O: [B, S, H*D]. No transpose copies on either side.
// pad: LDS row-padding in bf16 elements (pad sweep — NOTES r2: the +4
// layout still measured ~590K bank conflicts/dispatch).
}  // extern "C" (template below needs C++ linkage)

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

extern "C" int launch_attention_qkv_bf16_pad(const void* QKV, void* O,
                                             int B, int H, int S, int D,
                                             float scale, int pad,
                                             hipStream_t st) {
  if (!(S == 128 && D == 64)) return -1;
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
