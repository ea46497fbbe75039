// Order-preserving stream compaction + fused compare-and-compact + gather.
//
// MI355X-native replacement for DataFusion's FilterExec on the `sql`
// processor's hot path (reference processor/sql.rs:107-146). Two-pass
// block-count → scan → scatter keeps row order without global atomics;
// the scan between passes is one tiny device cumsum done by the caller.
#include "common.h"

// items per block: 256 threads × 4 = 1024 rows — small batches (8192 rows)
// still fill ≥8 blocks; big batches scale to thousands of workgroups.
#define FILTER_BLOCK 256
#define FILTER_IPT 4
#define FILTER_TILE (FILTER_BLOCK * FILTER_IPT)

enum CmpOp { LT = 0, LE = 1, GT = 2, GE = 3, EQ = 4, NE = 5 };

template <typename T>
DEV_INLINE bool cmp_apply(T v, int op, T s) {
  switch (op) {
    case LT: return v < s;
    case LE: return v <= s;
    case GT: return v > s;
    case GE: return v >= s;
    case EQ: return v == s;
    default: return v != s;
  }
}

// ---- pass 1: per-block count of matching rows -------------------------------
template <typename T>
__global__ void filter_count_kernel(const T* __restrict__ col, int64_t n,
                                    int op, T scalar,
                                    int32_t* __restrict__ block_counts) {
  int64_t base = (int64_t)blockIdx.x * FILTER_TILE;
  int cnt = 0;
#pragma unroll
  for (int i = 0; i < FILTER_IPT; ++i) {
    int64_t idx = base + threadIdx.x + i * FILTER_BLOCK;
    if (idx < n && cmp_apply(col[idx], op, scalar)) ++cnt;
  }
  // wave reduce then LDS across the 4 waves
  __shared__ int wsum[FILTER_BLOCK / WAVE];
  int v = cnt;
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if ((threadIdx.x & 63) == 0) wsum[threadIdx.x >> 6] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    int total = 0;
#pragma unroll
    for (int w = 0; w < FILTER_BLOCK / WAVE; ++w) total += wsum[w];
    block_counts[blockIdx.x] = total;
  }
}

// ---- pass 2: scatter matching row indices at block offsets ------------------
template <typename T>
__global__ void filter_scatter_kernel(const T* __restrict__ col, int64_t n,
                                      int op, T scalar,
                                      const int32_t* __restrict__ block_offsets,
                                      int32_t* __restrict__ out_idx) {
  int64_t base = (int64_t)blockIdx.x * FILTER_TILE;
  __shared__ int wave_base[FILTER_BLOCK / WAVE + 1];
  int write = block_offsets[blockIdx.x];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  // process the tile in FILTER_IPT ordered strips so output stays row-ordered
#pragma unroll
  for (int i = 0; i < FILTER_IPT; ++i) {
    int64_t idx = base + i * FILTER_BLOCK + threadIdx.x;
    bool pred = (idx < n) && cmp_apply(col[idx], op, scalar);
    uint64_t ballot = __ballot(pred);
    int rank = __popcll(ballot & lanemask_lt());
    int wave_total = __popcll(ballot);
    if (lane == 0) wave_base[wid + 1] = wave_total;
    __syncthreads();
    if (threadIdx.x == 0) {
      wave_base[0] = write;
      for (int w = 1; w <= FILTER_BLOCK / WAVE; ++w)
        wave_base[w] += wave_base[w - 1];
    }
    __syncthreads();
    if (pred) out_idx[wave_base[wid] + rank] = (int32_t)idx;
    write = wave_base[FILTER_BLOCK / WAVE];
    __syncthreads();
  }
}

// ---- mask (bool) variants ---------------------------------------------------
__global__ void mask_count_kernel(const bool* __restrict__ mask, int64_t n,
                                  int32_t* __restrict__ block_counts) {
  int64_t base = (int64_t)blockIdx.x * FILTER_TILE;
  int cnt = 0;
#pragma unroll
  for (int i = 0; i < FILTER_IPT; ++i) {
    int64_t idx = base + threadIdx.x + i * FILTER_BLOCK;
    if (idx < n && mask[idx]) ++cnt;
  }
  __shared__ int wsum[FILTER_BLOCK / WAVE];
  int v = cnt;
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if ((threadIdx.x & 63) == 0) wsum[threadIdx.x >> 6] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    int total = 0;
#pragma unroll
    for (int w = 0; w < FILTER_BLOCK / WAVE; ++w) total += wsum[w];
    block_counts[blockIdx.x] = total;
  }
}

__global__ void mask_scatter_kernel(const bool* __restrict__ mask, int64_t n,
                                    const int32_t* __restrict__ block_offsets,
                                    int32_t* __restrict__ out_idx) {
  int64_t base = (int64_t)blockIdx.x * FILTER_TILE;
  __shared__ int wave_base[FILTER_BLOCK / WAVE + 1];
  int write = block_offsets[blockIdx.x];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int i = 0; i < FILTER_IPT; ++i) {
    int64_t idx = base + i * FILTER_BLOCK + threadIdx.x;
    bool pred = (idx < n) && mask[idx];
    uint64_t ballot = __ballot(pred);
    int rank = __popcll(ballot & lanemask_lt());
    int wave_total = __popcll(ballot);
    if (lane == 0) wave_base[wid + 1] = wave_total;
    __syncthreads();
    if (threadIdx.x == 0) {
      wave_base[0] = write;
      for (int w = 1; w <= FILTER_BLOCK / WAVE; ++w)
        wave_base[w] += wave_base[w - 1];
    }
    __syncthreads();
    if (pred) out_idx[wave_base[wid] + rank] = (int32_t)idx;
    write = wave_base[FILTER_BLOCK / WAVE];
    __syncthreads();
  }
}

// ---- gather -----------------------------------------------------------------
// Coalesced row gather by element size; one thread per output row × chunk.
template <typename T>
__global__ void gather_kernel(const T* __restrict__ src,
                              const int32_t* __restrict__ idx, int64_t m,
                              T* __restrict__ dst) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < m; i += stride) dst[i] = src[idx[i]];
}

// ---- host-visible launchers (called from bindings.cpp) ----------------------
extern "C" {

int filter_grid(int64_t n) { return (int)((n + FILTER_TILE - 1) / FILTER_TILE); }

void launch_filter_count_f32(const float* col, int64_t n, int op, float s,
                             int32_t* counts, hipStream_t st) {
  filter_count_kernel<float><<<filter_grid(n), FILTER_BLOCK, 0, st>>>(
      col, n, op, s, counts);
}
void launch_filter_scatter_f32(const float* col, int64_t n, int op, float s,
                               const int32_t* offs, int32_t* out,
                               hipStream_t st) {
  filter_scatter_kernel<float><<<filter_grid(n), FILTER_BLOCK, 0, st>>>(
      col, n, op, s, offs, out);
}
void launch_filter_count_i64(const int64_t* col, int64_t n, int op, int64_t s,
                             int32_t* counts, hipStream_t st) {
  filter_count_kernel<int64_t><<<filter_grid(n), FILTER_BLOCK, 0, st>>>(
      col, n, op, s, counts);
}
void launch_filter_scatter_i64(const int64_t* col, int64_t n, int op,
                               int64_t s, const int32_t* offs, int32_t* out,
                               hipStream_t st) {
  filter_scatter_kernel<int64_t><<<filter_grid(n), FILTER_BLOCK, 0, st>>>(
      col, n, op, s, offs, out);
}
void launch_mask_count(const bool* mask, int64_t n, int32_t* counts,
                       hipStream_t st) {
  mask_count_kernel<<<filter_grid(n), FILTER_BLOCK, 0, st>>>(mask, n, counts);
}
void launch_mask_scatter(const bool* mask, int64_t n, const int32_t* offs,
                         int32_t* out, hipStream_t st) {
  mask_scatter_kernel<<<filter_grid(n), FILTER_BLOCK, 0, st>>>(
      mask, n, offs, out);
}

void launch_gather(const void* src, const int32_t* idx, int64_t m,
                   void* dst, int elem_size, hipStream_t st) {
  int grid = (int)((m + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid == 0) return;
  switch (elem_size) {
    case 1: gather_kernel<uint8_t><<<grid, 256, 0, st>>>(
        (const uint8_t*)src, idx, m, (uint8_t*)dst); break;
    case 2: gather_kernel<uint16_t><<<grid, 256, 0, st>>>(
        (const uint16_t*)src, idx, m, (uint16_t*)dst); break;
    case 4: gather_kernel<uint32_t><<<grid, 256, 0, st>>>(
        (const uint32_t*)src, idx, m, (uint32_t*)dst); break;
    default: gather_kernel<uint64_t><<<grid, 256, 0, st>>>(
        (const uint64_t*)src, idx, m, (uint64_t*)dst); break;
  }
}

}  // extern "C"

// ---- multi-column gather: one launch for a whole batch ----------------------
#define GATHER_MAX_COLS 32
struct GatherSpec {
  int ncols;
  const void* src[GATHER_MAX_COLS];
  void* dst[GATHER_MAX_COLS];
  int esz[GATHER_MAX_COLS];
};

__global__ void gather_multi_kernel(GatherSpec spec,
                                    const int32_t* __restrict__ idx,
                                    int64_t m) {
  int64_t total = (int64_t)spec.ncols * m;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int c = (int)(i / m);
    int64_t j = i - (int64_t)c * m;
    int64_t s = idx[j];
    switch (spec.esz[c]) {
      case 1: ((uint8_t*)spec.dst[c])[j] = ((const uint8_t*)spec.src[c])[s];
              break;
      case 2: ((uint16_t*)spec.dst[c])[j] = ((const uint16_t*)spec.src[c])[s];
              break;
      case 4: ((uint32_t*)spec.dst[c])[j] = ((const uint32_t*)spec.src[c])[s];
              break;
      default: ((uint64_t*)spec.dst[c])[j] = ((const uint64_t*)spec.src[c])[s];
               break;
    }
  }
}

extern "C" void launch_gather_multi(int ncols, const void** src, void** dst,
                                    const int* esz, const int32_t* idx,
                                    int64_t m, hipStream_t st) {
  GatherSpec spec{};
  spec.ncols = ncols;
  for (int c = 0; c < ncols; ++c) {
    spec.src[c] = src[c];
    spec.dst[c] = dst[c];
    spec.esz[c] = esz[c];
  }
  int64_t total = (int64_t)ncols * m;
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
  if (grid < 1) return;
  gather_multi_kernel<<<grid, 256, 0, st>>>(spec, idx, m);
}

// ---- dyn-count multi-gather: hipGraph-capturable compaction tail ------------
// Same as gather_multi_kernel but the row count lives in DEVICE memory
// (written by the scan), so the whole filter→gather chain captures into a
// hipGraph with no host sync. Launched with a grid sized for the batch cap;
// threads past *count_dev exit.
__global__ void gather_multi_dyn_kernel(GatherSpec spec,
                                        const int32_t* __restrict__ idx,
                                        const int32_t* __restrict__ count_dev,
                                        int64_t cap) {
  int64_t m = (int64_t)*count_dev;
  int64_t total = (int64_t)spec.ncols * m;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int c = (int)(i / m);
    int64_t j = i - (int64_t)c * m;
    int64_t s = idx[j];
    switch (spec.esz[c]) {
      case 1: ((uint8_t*)spec.dst[c])[j] = ((const uint8_t*)spec.src[c])[s];
              break;
      case 2: ((uint16_t*)spec.dst[c])[j] = ((const uint16_t*)spec.src[c])[s];
              break;
      case 4: ((uint32_t*)spec.dst[c])[j] = ((const uint32_t*)spec.src[c])[s];
              break;
      default: ((uint64_t*)spec.dst[c])[j] = ((const uint64_t*)spec.src[c])[s];
               break;
    }
  }
}

extern "C" void launch_gather_multi_dyn(int ncols, const void** src,
                                        void** dst, const int* esz,
                                        const int32_t* idx,
                                        const int32_t* count_dev, int64_t cap,
                                        hipStream_t st) {
  GatherSpec spec{};
  spec.ncols = ncols;
  for (int c = 0; c < ncols; ++c) {
    spec.src[c] = src[c];
    spec.dst[c] = dst[c];
    spec.esz[c] = esz[c];
  }
  int64_t total = (int64_t)ncols * cap;
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
  if (grid < 1) return;
  gather_multi_dyn_kernel<<<grid, 256, 0, st>>>(spec, idx, count_dev, cap);
}

// ---- per-row bytes hash (binary columns: group-by / repartition keys) -------
// 64-bit FNV-1a over each row's byte slice; one thread per row. Used to
// dictionary-encode string keys on-device (collision probability ~n²/2^65 —
// negligible at stream batch sizes; exact verification is the CPU path).
__global__ void bytes_hash_kernel(const uint8_t* __restrict__ data,
                                  const int64_t* __restrict__ offsets,
                                  int64_t n, int64_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t h = 0xcbf29ce484222325ull;
    for (int64_t p = offsets[i]; p < offsets[i + 1]; ++p) {
      h ^= data[p];
      h *= 0x100000001b3ull;
    }
    out[i] = (int64_t)h;
  }
}

extern "C" void launch_bytes_hash(const uint8_t* data, const int64_t* offsets,
                                  int64_t n, int64_t* out, hipStream_t st) {
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  bytes_hash_kernel<<<grid, 256, 0, st>>>(data, offsets, n, out);
}

// ------------------------------------------------------------------ prefix sum
// Exclusive prefix sum of an int32 array into int64 offsets — the device-side
// cumsum for string-column copy-out (json/proto decoders) and any
// offsets-from-lengths step. Two passes: per-block sums, then (after a tiny
// torch cumsum over <=2048 block sums) an in-block scan + add. Replaces
// torch's innermost-dim scan kernel, which is ~600 µs at n=262K vs ~15 µs
// for this pair.
#define SCAN_BLOCK 256
#define SCAN_IPT 8
#define SCAN_TILE (SCAN_BLOCK * SCAN_IPT)

__global__ void scan_partials_kernel(const int32_t* __restrict__ in, int64_t n,
                                     int64_t* __restrict__ partials) {
  int64_t base = (int64_t)blockIdx.x * SCAN_TILE;
  int64_t local = 0;
#pragma unroll
  for (int k = 0; k < SCAN_IPT; ++k) {
    int64_t i = base + threadIdx.x * SCAN_IPT + k;
    if (i < n) local += in[i];
  }
  __shared__ int64_t lds[SCAN_BLOCK];
  lds[threadIdx.x] = local;
  __syncthreads();
  // tree reduce
  for (int s = SCAN_BLOCK / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) lds[threadIdx.x] += lds[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) partials[blockIdx.x] = lds[0];
}

__global__ void scan_write_kernel(const int32_t* __restrict__ in, int64_t n,
                                  const int64_t* __restrict__ block_offs,
                                  int64_t* __restrict__ out) {
  int64_t base = (int64_t)blockIdx.x * SCAN_TILE;
  int32_t vals[SCAN_IPT];
  int64_t local = 0;
#pragma unroll
  for (int k = 0; k < SCAN_IPT; ++k) {
    int64_t i = base + threadIdx.x * SCAN_IPT + k;
    vals[k] = i < n ? in[i] : 0;
    local += vals[k];
  }
  __shared__ int64_t lds[SCAN_BLOCK];
  lds[threadIdx.x] = local;
  __syncthreads();
  // Hillis-Steele inclusive scan over per-thread sums
  for (int s = 1; s < SCAN_BLOCK; s <<= 1) {
    int64_t add = threadIdx.x >= s ? lds[threadIdx.x - s] : 0;
    __syncthreads();
    lds[threadIdx.x] += add;
    __syncthreads();
  }
  int64_t run = block_offs[blockIdx.x] +
                (threadIdx.x ? lds[threadIdx.x - 1] : 0);
#pragma unroll
  for (int k = 0; k < SCAN_IPT; ++k) {
    int64_t i = base + threadIdx.x * SCAN_IPT + k;
    if (i < n) {
      out[i] = run;  // exclusive
      run += vals[k];
    }
  }
}

// single-block exclusive scan of ≤1024 int64 block sums → block offsets +
// grand total (replaces the torch cumsum + sub + copy chain between the
// two scan passes — 3-4 launches → 1)
__global__ void scan_boffs64_kernel(const int64_t* __restrict__ partials,
                                    int nb, int64_t* __restrict__ boffs,
                                    int64_t* __restrict__ total_out) {
  __shared__ int64_t lds[1024];
  int tid = threadIdx.x;
  int64_t v = tid < nb ? partials[tid] : 0;
  lds[tid] = v;
  __syncthreads();
  for (int s = 1; s < 1024; s <<= 1) {
    int64_t add = tid >= s ? lds[tid - s] : 0;
    __syncthreads();
    lds[tid] += add;
    __syncthreads();
  }
  if (tid < nb) boffs[tid] = lds[tid] - v;
  if (tid == nb - 1) *total_out = lds[tid];
}

extern "C" void launch_scan_boffs64(const int64_t* partials, int nb,
                                    int64_t* boffs, int64_t* total_out,
                                    hipStream_t st) {
  scan_boffs64_kernel<<<1, 1024, 0, st>>>(partials, nb, boffs, total_out);
}

extern "C" int scan_grid(int64_t n) {
  return (int)((n + SCAN_TILE - 1) / SCAN_TILE);
}

extern "C" void launch_scan_partials(const int32_t* in, int64_t n,
                                     int64_t* partials, hipStream_t st) {
  scan_partials_kernel<<<scan_grid(n), SCAN_BLOCK, 0, st>>>(in, n, partials);
}

extern "C" void launch_scan_write(const int32_t* in, int64_t n,
                                  const int64_t* block_offs, int64_t* out,
                                  hipStream_t st) {
  scan_write_kernel<<<scan_grid(n), SCAN_BLOCK, 0, st>>>(in, n, block_offs,
                                                         out);
}

// ------------------------------------------------------------------ LIKE match
// SQL LIKE on a binary column entirely on-device: mode 0 = contains (%x%),
// 1 = prefix (x%), 2 = suffix (%x), 3 = equals. Needle passed by value in
// the kernarg segment (bounded, like JsonSpec).
#define MATCH_MAX_NEEDLE 64

struct MatchNeedle {
  uint8_t bytes[MATCH_MAX_NEEDLE];
  int len;
  int mode;
};

__global__ void bytes_match_kernel(const uint8_t* __restrict__ data,
                                   const int64_t* __restrict__ offsets,
                                   int64_t n, MatchNeedle nd,
                                   bool* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t s = offsets[i], e = offsets[i + 1];
    const int64_t len = e - s;
    bool m = false;
    if (nd.len == 0) {
      m = (nd.mode == 3) ? (len == 0) : true;
    } else if (len >= nd.len) {
      if (nd.mode == 1 || nd.mode == 3) {        // prefix / equals
        m = (nd.mode == 1 || len == nd.len);
        for (int k = 0; m && k < nd.len; ++k)
          m = data[s + k] == nd.bytes[k];
      } else if (nd.mode == 2) {                 // suffix
        m = true;
        for (int k = 0; m && k < nd.len; ++k)
          m = data[e - nd.len + k] == nd.bytes[k];
      } else {                                   // contains
        for (int64_t p = s; !m && p + nd.len <= e; ++p) {
          bool eq = true;
          for (int k = 0; eq && k < nd.len; ++k)
            eq = data[p + k] == nd.bytes[k];
          m = eq;
        }
      }
    }
    out[i] = m;
  }
}

extern "C" void launch_bytes_match(const uint8_t* data, const int64_t* offsets,
                                   int64_t n, const uint8_t* needle, int nl,
                                   int mode, bool* out, hipStream_t st) {
  MatchNeedle nd{};
  nd.len = nl > MATCH_MAX_NEEDLE ? MATCH_MAX_NEEDLE : nl;
  nd.mode = mode;
  for (int k = 0; k < nd.len; ++k) nd.bytes[k] = needle[k];
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  bytes_match_kernel<<<grid, 256, 0, st>>>(data, offsets, n, nd, out);
}
