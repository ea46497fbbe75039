// Row-wise bf16 ops for the inference processor: LayerNorm (with optional
// fused residual add) and row softmax. Memory-bound → vectorized short4/short8
// loads per guide Guideline 13; one block per row, wave+LDS reductions.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(8))) __bf16 vbf16x8;

#define ROW_THREADS 256

DEV_INLINE float block_reduce_sum(float v, float* lds) {
  v = wave_reduce_sum(v);
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < ROW_THREADS / WAVE; ++w) total += lds[w];
  __syncthreads();
  return total;
}

DEV_INLINE float block_reduce_max(float v, float* lds) {
  v = wave_reduce_max(v);
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds[wid] = v;
  __syncthreads();
  float total = -INFINITY;
#pragma unroll
  for (int w = 0; w < ROW_THREADS / WAVE; ++w) total = fmaxf(total, lds[w]);
  __syncthreads();
  return total;
}

// ---- LayerNorm ---------------------------------------------------------------
// x:[rows,n] bf16 (+ optional residual), gamma/beta:[n] f32 → out bf16.
// n must be a multiple of 8 (bf16x8 vector path).
template <bool ADD_RESIDUAL>
__global__ __launch_bounds__(ROW_THREADS)
void layernorm_bf16_kernel(const __bf16* __restrict__ x,
                           const __bf16* __restrict__ residual,
                           const float* __restrict__ gamma,
                           const float* __restrict__ beta,
                           __bf16* __restrict__ out,
                           __bf16* __restrict__ resid_out,  // optional x+res
                           int64_t rows, int n, float eps) {
  __shared__ float red[ROW_THREADS / WAVE];
  const int nv = n / 8;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const vbf16x8* xr = (const vbf16x8*)(x + row * n);
    const vbf16x8* rr =
        ADD_RESIDUAL ? (const vbf16x8*)(residual + row * n) : nullptr;
    float sum = 0.f, sq = 0.f;
    for (int i = threadIdx.x; i < nv; i += ROW_THREADS) {
      vbf16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        if (ADD_RESIDUAL) f += (float)rr[i][j];
        sum += f;
        sq += f * f;
      }
    }
    float mean = block_reduce_sum(sum, red) / n;
    float var = block_reduce_sum(sq, red) / n - mean * mean;
    float rstd = rsqrtf(var + eps);
    vbf16x8* orow = (vbf16x8*)(out + row * n);
    vbf16x8* resrow =
        (ADD_RESIDUAL && resid_out) ? (vbf16x8*)(resid_out + row * n) : nullptr;
    for (int i = threadIdx.x; i < nv; i += ROW_THREADS) {
      vbf16x8 v = xr[i];
      vbf16x8 o, rsum;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        if (ADD_RESIDUAL) f += (float)rr[i][j];
        if (ADD_RESIDUAL && resrow) rsum[j] = (__bf16)f;
        int c = i * 8 + j;
        o[j] = (__bf16)((f - mean) * rstd * gamma[c] + beta[c]);
      }
      orow[i] = o;
      if (ADD_RESIDUAL && resrow) resrow[i] = rsum;
    }
    __syncthreads();
  }
}

// wave-per-row LayerNorm: at BERT width (n=768 → 96 bf16x8 vectors) the
// block-per-row kernel leaves 160 of 256 threads idle and pays two block
// barriers per row; a wave per row uses shuffle-only reductions and packs
// 4 rows per block (measured ~2× at 8192×768 — profiles r2).
template <bool ADD_RESIDUAL>
__global__ __launch_bounds__(ROW_THREADS)
void layernorm_bf16_wave_kernel(const __bf16* __restrict__ x,
                                const __bf16* __restrict__ residual,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                __bf16* __restrict__ out,
                                __bf16* __restrict__ resid_out,
                                int64_t rows, int n, float eps) {
  const int nv = n / 8;
  const int lane = threadIdx.x & 63;
  int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int64_t wstride = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const bool cached = nv <= 2 * WAVE;  // lane holds <=2 vectors in regs
  for (int64_t row = wave; row < rows; row += wstride) {
    const vbf16x8* xr = (const vbf16x8*)(x + row * n);
    const vbf16x8* rr =
        ADD_RESIDUAL ? (const vbf16x8*)(residual + row * n) : nullptr;
    vbf16x8* orow = (vbf16x8*)(out + row * n);
    vbf16x8* resrow = (ADD_RESIDUAL && resid_out)
                          ? (vbf16x8*)(resid_out + row * n)
                          : nullptr;
    float sum = 0.f, sq = 0.f;
    float c0[16];  // row cache: the (x+residual) floats, read-once
    if (cached) {
#pragma unroll
      for (int t = 0; t < 2; ++t) {
        int i = lane + t * WAVE;
        if (i < nv) {
          vbf16x8 v = xr[i];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float f = (float)v[j];
            if (ADD_RESIDUAL) f += (float)rr[i][j];
            c0[t * 8 + j] = f;
            sum += f;
            sq += f * f;
          }
        }
      }
    } else {
      for (int i = lane; i < nv; i += WAVE) {
        vbf16x8 v = xr[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v[j];
          if (ADD_RESIDUAL) f += (float)rr[i][j];
          sum += f;
          sq += f * f;
        }
      }
    }
    sum = wave_reduce_sum(sum);
    sq = wave_reduce_sum(sq);
    sum = __shfl(sum, 0, 64);  // broadcast (reduce leaves lane 0 complete)
    sq = __shfl(sq, 0, 64);
    float mean = sum / n;
    float var = sq / n - mean * mean;
    float rstd = rsqrtf(var + eps);
    if (cached) {
#pragma unroll
      for (int t = 0; t < 2; ++t) {
        int i = lane + t * WAVE;
        if (i < nv) {
          vbf16x8 o, rsum;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float f = c0[t * 8 + j];
            if (ADD_RESIDUAL && resrow) rsum[j] = (__bf16)f;
            int c = i * 8 + j;
            o[j] = (__bf16)((f - mean) * rstd * gamma[c] + beta[c]);
          }
          orow[i] = o;
          if (ADD_RESIDUAL && resrow) resrow[i] = rsum;
        }
      }
    } else {
      for (int i = lane; i < nv; i += WAVE) {
        vbf16x8 v = xr[i];
        vbf16x8 o, rsum;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v[j];
          if (ADD_RESIDUAL) f += (float)rr[i][j];
          if (ADD_RESIDUAL && resrow) rsum[j] = (__bf16)f;
          int c = i * 8 + j;
          o[j] = (__bf16)((f - mean) * rstd * gamma[c] + beta[c]);
        }
        orow[i] = o;
        if (ADD_RESIDUAL && resrow) resrow[i] = rsum;
      }
    }
  }
}

// ---- Softmax -----------------------------------------------------------------
// Row softmax over the last dim (n multiple of 8), numerically stable.
__global__ __launch_bounds__(ROW_THREADS)
void softmax_bf16_kernel(const __bf16* __restrict__ x, __bf16* __restrict__ out,
                         int64_t rows, int n, float scale) {
  __shared__ float red[ROW_THREADS / WAVE];
  const int nv = n / 8;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const vbf16x8* xr = (const vbf16x8*)(x + row * n);
    float m = -INFINITY;
    for (int i = threadIdx.x; i < nv; i += ROW_THREADS) {
      vbf16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, (float)v[j] * scale);
    }
    m = block_reduce_max(m, red);
    float sum = 0.f;
    for (int i = threadIdx.x; i < nv; i += ROW_THREADS) {
      vbf16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) sum += __expf((float)v[j] * scale - m);
    }
    sum = block_reduce_sum(sum, red);
    float inv = 1.f / sum;
    vbf16x8* orow = (vbf16x8*)(out + row * n);
    for (int i = threadIdx.x; i < nv; i += ROW_THREADS) {
      vbf16x8 v = xr[i];
      vbf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (__bf16)(__expf((float)v[j] * scale - m) * inv);
      orow[i] = o;
    }
    __syncthreads();
  }
}

// ---- elementwise bias+activation (fallback when not fused into GEMM) ---------
__global__ void bias_act_bf16_kernel(const __bf16* __restrict__ x,
                                     const float* __restrict__ bias,
                                     __bf16* __restrict__ out, int64_t rows,
                                     int n, int act) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = rows * n;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    float v = (float)x[i] + (bias ? bias[i % n] : 0.f);
    switch (act) {
      case 1: v = fmaxf(v, 0.f); break;
      case 2: {
        float c = 0.7978845608028654f * (v + 0.044715f * v * v * v);
        c = fminf(fmaxf(c, -10.f), 10.f);
        float e = __expf(2.f * c);
        v = 0.5f * v * (1.f + (e - 1.f) / (e + 1.f));
        break;
      }
      case 3: v = v / (1.f + __expf(-v)); break;
    }
    out[i] = (__bf16)v;
  }
}

extern "C" {

void launch_layernorm_bf16(const void* x, const void* residual,
                           const float* gamma, const float* beta, void* out,
                           void* resid_out, int64_t rows, int n, float eps,
                           hipStream_t st) {
  if (rows < 1) return;
  if (n <= 2048) {  // wave-per-row: shuffle-only reductions, 4 rows/block
    int64_t waves = rows;
    int grid = (int)((waves * 64 + ROW_THREADS - 1) / ROW_THREADS);
    if (grid > 4096) grid = 4096;
    if (residual)
      layernorm_bf16_wave_kernel<true><<<grid, ROW_THREADS, 0, st>>>(
          (const __bf16*)x, (const __bf16*)residual, gamma, beta,
          (__bf16*)out, (__bf16*)resid_out, rows, n, eps);
    else
      layernorm_bf16_wave_kernel<false><<<grid, ROW_THREADS, 0, st>>>(
          (const __bf16*)x, nullptr, gamma, beta, (__bf16*)out, nullptr,
          rows, n, eps);
    return;
  }
  int grid = rows < 2048 ? (int)rows : 2048;
  if (grid < 1) return;
  if (residual)
    layernorm_bf16_kernel<true><<<grid, ROW_THREADS, 0, st>>>(
        (const __bf16*)x, (const __bf16*)residual, gamma, beta, (__bf16*)out,
        (__bf16*)resid_out, rows, n, eps);
  else
    layernorm_bf16_kernel<false><<<grid, ROW_THREADS, 0, st>>>(
        (const __bf16*)x, nullptr, gamma, beta, (__bf16*)out, nullptr, rows,
        n, eps);
}

void launch_softmax_bf16(const void* x, void* out, int64_t rows, int n,
                         float scale, hipStream_t st) {
  int grid = rows < 4096 ? (int)rows : 4096;
  if (grid < 1) return;
  softmax_bf16_kernel<<<grid, ROW_THREADS, 0, st>>>(
      (const __bf16*)x, (__bf16*)out, rows, n, scale);
}

void launch_bias_act_bf16(const void* x, const float* bias, void* out,
                          int64_t rows, int n, int act, hipStream_t st) {
  int64_t total = rows * n;
  int grid = (int)((total + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  bias_act_bf16_kernel<<<grid, 256, 0, st>>>(
      (const __bf16*)x, bias, (__bf16*)out, rows, n, act);
}

}  // extern "C"
