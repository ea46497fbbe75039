// Device radix argsort — ORDER BY / sort physical operator
// ("sort/order-by, distinct → device radix sort kernels").
//
// Stable LSD radix sort over 4-bit digits producing a permutation (int32
// indices). Keys: uint32 (f32 via monotone flip, int32 via bias) = 8 passes;
// uint64 (int64 via bias) = 16 passes. Each pass: per-block histogram →
// device exclusive scan (caller, one cumsum) → order-preserving scatter with
// per-wave ballot ranking (same two-pass pattern as filter.hip).
#include "common.h"

#define RS_BLOCK 256
#define RS_IPT 4
#define RS_TILE (RS_BLOCK * RS_IPT)
#define RS_BINS 16

template <typename K>
DEV_INLINE int digit_of(K key, int shift) {
  return (int)((key >> shift) & (K)(RS_BINS - 1));
}

// ---- pass 1: per-block digit histogram → hist[bin * nblocks + block] --------
template <typename K>
__global__ void radix_hist_kernel(const K* __restrict__ keys,
                                  const int32_t* __restrict__ idx_in,
                                  int64_t n, int shift,
                                  int32_t* __restrict__ hist, int nblocks) {
  __shared__ int local[RS_BINS];
  for (int b = threadIdx.x; b < RS_BINS; b += RS_BLOCK) local[b] = 0;
  __syncthreads();
  int64_t base = (int64_t)blockIdx.x * RS_TILE;
#pragma unroll
  for (int i = 0; i < RS_IPT; ++i) {
    int64_t j = base + threadIdx.x + i * RS_BLOCK;
    if (j < n) {
      K k = keys[idx_in[j]];
      atomicAdd(&local[digit_of(k, shift)], 1);
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < RS_BINS; b += RS_BLOCK)
    hist[(int64_t)b * nblocks + blockIdx.x] = local[b];
}

// ---- pass 2: stable scatter at scanned offsets -------------------------------
template <typename K>
__global__ void radix_scatter_kernel(const K* __restrict__ keys,
                                     const int32_t* __restrict__ idx_in,
                                     int64_t n, int shift,
                                     const int32_t* __restrict__ offsets,
                                     int nblocks,
                                     int32_t* __restrict__ idx_out) {
  // write cursor per bin for this block (starts at the scanned offset)
  __shared__ int cursor[RS_BINS];
  __shared__ int wave_cnt[RS_BINS][RS_BLOCK / WAVE + 1];
  for (int b = threadIdx.x; b < RS_BINS; b += RS_BLOCK)
    cursor[b] = offsets[(int64_t)b * nblocks + blockIdx.x];
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  int64_t base = (int64_t)blockIdx.x * RS_TILE;
#pragma unroll
  for (int i = 0; i < RS_IPT; ++i) {
    int64_t j = base + i * RS_BLOCK + threadIdx.x;
    bool valid = j < n;
    int32_t src = valid ? idx_in[j] : 0;
    int d = valid ? digit_of(keys[src], shift) : -1;
    // per-wave, per-bin ballot ranking (stable within the strip)
    int rank = 0;
    uint64_t mask_mine = 0;
#pragma unroll
    for (int b = 0; b < RS_BINS; ++b) {
      uint64_t m = __ballot(d == b);
      if (d == b) {
        mask_mine = m;
        rank = __popcll(m & lanemask_lt());
      }
      if (lane == 0) wave_cnt[b][wid + 1] = __popcll(m);
    }
    __syncthreads();
    if (threadIdx.x < RS_BINS) {
      int b = threadIdx.x;
      wave_cnt[b][0] = cursor[b];
      for (int w = 1; w <= RS_BLOCK / WAVE; ++w)
        wave_cnt[b][w] += wave_cnt[b][w - 1];
      cursor[b] = wave_cnt[b][RS_BLOCK / WAVE];
    }
    __syncthreads();
    if (valid) idx_out[wave_cnt[d][wid] + rank] = src;
    __syncthreads();
  }
}

extern "C" {

static int rs_grid(int64_t n) {
  return (int)((n + RS_TILE - 1) / RS_TILE);
}

int radix_sort_nblocks(int64_t n) { return rs_grid(n); }

void launch_radix_hist_u32(const uint32_t* keys, const int32_t* idx_in,
                           int64_t n, int shift, int32_t* hist, int nblocks,
                           hipStream_t st) {
  radix_hist_kernel<uint32_t><<<rs_grid(n), RS_BLOCK, 0, st>>>(
      keys, idx_in, n, shift, hist, nblocks);
}
void launch_radix_scatter_u32(const uint32_t* keys, const int32_t* idx_in,
                              int64_t n, int shift, const int32_t* offsets,
                              int nblocks, int32_t* idx_out, hipStream_t st) {
  radix_scatter_kernel<uint32_t><<<rs_grid(n), RS_BLOCK, 0, st>>>(
      keys, idx_in, n, shift, offsets, nblocks, idx_out);
}
void launch_radix_hist_u64(const uint64_t* keys, const int32_t* idx_in,
                           int64_t n, int shift, int32_t* hist, int nblocks,
                           hipStream_t st) {
  radix_hist_kernel<uint64_t><<<rs_grid(n), RS_BLOCK, 0, st>>>(
      keys, idx_in, n, shift, hist, nblocks);
}
void launch_radix_scatter_u64(const uint64_t* keys, const int32_t* idx_in,
                              int64_t n, int shift, const int32_t* offsets,
                              int nblocks, int32_t* idx_out, hipStream_t st) {
  radix_scatter_kernel<uint64_t><<<rs_grid(n), RS_BLOCK, 0, st>>>(
      keys, idx_in, n, shift, offsets, nblocks, idx_out);
}

// key transforms: float → monotone uint32; int → biased unsigned
__global__ void f32_to_ordered_u32(const float* __restrict__ in,
                                   uint32_t* __restrict__ out, int64_t n,
                                   int descending) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint32_t u = float_flip(in[i]);
    out[i] = descending ? ~u : u;
  }
}
__global__ void i64_to_ordered_u64(const int64_t* __restrict__ in,
                                   uint64_t* __restrict__ out, int64_t n,
                                   int descending) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t u = (uint64_t)in[i] ^ 0x8000000000000000ull;
    out[i] = descending ? ~u : u;
  }
}

void launch_f32_to_ordered(const float* in, uint32_t* out, int64_t n,
                           int descending, hipStream_t st) {
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  f32_to_ordered_u32<<<grid, 256, 0, st>>>(in, out, n, descending);
}
void launch_i64_to_ordered(const int64_t* in, uint64_t* out, int64_t n,
                           int descending, hipStream_t st) {
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  i64_to_ordered_u64<<<grid, 256, 0, st>>>(in, out, n, descending);
}

}  // extern "C"
