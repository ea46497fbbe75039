// Fused kernels for the flagship whole-step hipGraph
// (generate → filter → MLP; ops/stepgraph.py).
//
// The graphed step was still ~21 kernels: torch's rand/addcmul/randint,
// cumsum chain, stack/pad/cast, and a hipBLASLt GEMV each cost a ~4-5 µs
// launch slot even inside a graph. These kernels collapse that tail:
//
//   gen_fields      rand for ALL float columns + the int64 key column in one
//                   launch (stateless counter-hash RNG; the counter advances
//                   on-device so every graph replay re-randomizes)
//   scan_counts     single-block exclusive scan of the filter's block counts
//                   → offsets + device total (replaces cumsum+sub+copy)
//   featpack        gathered f32 columns → [n, K] bf16 padded MFMA operand
//                   (replaces stack + cast + pad)
//   gemv_bf16_f32   [M,K]·[K] + bias → f32 scores (replaces the final
//                   hipBLASLt N=1 GEMM + the bf16→f32 copy)
#include "common.h"

// counter-based RNG: mix64(ctr ^ element-id) — statistical quality is ample
// for synthetic benchmark data (uniform floats / uniform ints).
DEV_INLINE float u01(uint64_t h) {
  return (float)(h >> 40) * (1.0f / 16777216.0f);  // top 24 bits
}

struct GenSpec {
  float lo[32];
  float width[32];   // high - low
  int nf;            // float columns (<=32)
  int64_t key_lo;
  int64_t key_range; // 0 = no key column
};

// block [nf, n] float32 (rows are the columns) + key[n] int64.
// ctr[0] = replay counter (read by all, advanced by the last block to
// finish via the ctr[1] ticket — grid-wide single increment per launch).
__global__ void gen_fields_kernel(float* __restrict__ block,
                                  int64_t* __restrict__ key, int64_t n,
                                  GenSpec spec,
                                  unsigned long long* __restrict__ ctr) {
  uint64_t c = ctr[0];
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t base = (c << 32) ^ (uint64_t)i;
#pragma unroll 4
    for (int f = 0; f < spec.nf; ++f) {
      uint64_t h = mix64(base + ((uint64_t)(f + 1) << 52));
      block[(int64_t)f * n + i] = spec.lo[f] + spec.width[f] * u01(h);
    }
    if (spec.key_range > 0) {
      uint64_t h = mix64(base + (0x9E37ull << 48));
      key[i] = spec.key_lo + (int64_t)(h % (uint64_t)spec.key_range);
    }
  }
  // last block to arrive bumps the replay counter
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned long long t = atomicAdd(&ctr[1], 1ull);
    if (t == (unsigned long long)gridDim.x - 1) {
      ctr[1] = 0ull;
      ctr[0] = c + 1ull;
      __threadfence();
    }
  }
}

extern "C" void launch_gen_fields(float* block, int64_t* key, int64_t n,
                                  const float* lo, const float* width,
                                  int nf, int64_t key_lo, int64_t key_range,
                                  unsigned long long* ctr, hipStream_t st) {
  GenSpec spec{};
  spec.nf = nf > 32 ? 32 : nf;
  for (int f = 0; f < spec.nf; ++f) {
    spec.lo[f] = lo[f];
    spec.width[f] = width[f];
  }
  spec.key_lo = key_lo;
  spec.key_range = key_range;
  int grid = (int)((n + 255) / 256);
  if (grid > 1024) grid = 1024;
  if (grid < 1) return;
  gen_fields_kernel<<<grid, 256, 0, st>>>(block, key, n, spec, ctr);
}

// ---- single-block scan: counts[nb] → offs[nb] (exclusive) + total ----------
// nb is the filter's block count (batch/1024), ≤ 1024 here. One workgroup.
__global__ void scan_counts_kernel(const int32_t* __restrict__ counts, int nb,
                                   int32_t* __restrict__ offs,
                                   int32_t* __restrict__ total) {
  __shared__ int32_t lds[1024];
  int tid = threadIdx.x;
  int v = tid < nb ? counts[tid] : 0;
  lds[tid] = v;
  __syncthreads();
  // Hillis-Steele inclusive
  for (int s = 1; s < 1024; s <<= 1) {
    int add = tid >= s ? lds[tid - s] : 0;
    __syncthreads();
    lds[tid] += add;
    __syncthreads();
  }
  if (tid < nb) offs[tid] = lds[tid] - v;  // exclusive
  if (tid == nb - 1) *total = lds[tid];
}

extern "C" void launch_scan_counts(const int32_t* counts, int nb,
                                   int32_t* offs, int32_t* total,
                                   hipStream_t st) {
  scan_counts_kernel<<<1, 1024, 0, st>>>(counts, nb, offs, total);
}

// ---- whole-front fusion: generate + filter + compact + featpack ------------
// One persistent kernel replaces the gen_fields → filter count → scan →
// scatter → gather → featpack chain (5-6 launches, each a ~4-5 us slot at
// small grids). The RNG is a pure counter-hash, so phase C REGENERATES the
// surviving rows' values instead of re-reading a staged block — the
// generated batch never touches memory at all. One in-kernel grid barrier
// (monotonic agent-scope counter; launcher caps the grid at 256 blocks so
// all blocks are co-resident) orders the per-block counts before the
// compacted writes. Reference analog: the whole filter stage of
// stream/mod.rs:370-444 + DataFusion FilterExec as one device program.
struct GfpOuts {
  float* outs[32];
  int64_t* key_out;
  __bf16* feats;  // nullptr = no feature packing
  int32_t* count_out;
  int32_t* counts_ws;       // [grid]
  unsigned long long* bar;  // [1] monotonic barrier generation counter
  unsigned long long* ctr;  // [2] RNG replay counter + advance ticket
};

#define GFP_CH 1024  // MAX rows per block; the launcher shrinks the chunk
                     // so small batches still spread over >=32 blocks

__global__ __launch_bounds__(256)
void genfiltpack_kernel(GenSpec spec, int64_t n, int ch, int fidx, int op,
                        float scalar, int dpad, GfpOuts o) {
  const uint64_t c = o.ctr[0];
  const int b = blockIdx.x, grid = gridDim.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int64_t start = (int64_t)b * ch;
  __shared__ uint16_t loc[GFP_CH];
  __shared__ int wave_base[5];
  __shared__ int lcount_sh;
  __shared__ int base_sh;
  if (tid == 0) lcount_sh = 0;
  __syncthreads();
  // phase A: predicate on the filter field (recomputed from the counter
  // RNG), ordered in-block compaction of local row ids
  for (int s = 0; s < ch / 256; ++s) {
    int64_t idx = start + s * 256 + tid;
    bool pred = false;
    if (idx < n) {
      uint64_t rb = (c << 32) ^ (uint64_t)idx;
      uint64_t h = mix64(rb + ((uint64_t)(fidx + 1) << 52));
      float v = spec.lo[fidx] + spec.width[fidx] * u01(h);
      switch (op) {
        case 0: pred = v < scalar; break;
        case 1: pred = v <= scalar; break;
        case 2: pred = v > scalar; break;
        case 3: pred = v >= scalar; break;
        case 4: pred = v == scalar; break;
        default: pred = v != scalar; break;  // != NaN ⇒ keep-all
      }
    }
    uint64_t ballot = __ballot(pred);
    int rank = __popcll(ballot & lanemask_lt());
    if (lane == 0) wave_base[wid + 1] = __popcll(ballot);
    __syncthreads();
    if (tid == 0) {
      wave_base[0] = lcount_sh;
      for (int w = 1; w <= 4; ++w) wave_base[w] += wave_base[w - 1];
      lcount_sh = wave_base[4];
    }
    __syncthreads();
    if (pred) loc[wave_base[wid] + rank] = (uint16_t)(s * 256 + tid);
    __syncthreads();
  }
  const int lcount = lcount_sh;
  if (tid == 0)
    __hip_atomic_store(&o.counts_ws[b], lcount, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_AGENT);
  // grid barrier: generation = floor(old/grid)+1 — monotonic, so graph
  // replays need no reset and stragglers can never miss the flag
  if (tid == 0) {
    unsigned long long old = atomicAdd(o.bar, 1ull);
    unsigned long long target =
        (old / grid + 1) * (unsigned long long)grid;
    while (__hip_atomic_load(o.bar, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_AGENT) < target)
      __builtin_amdgcn_s_sleep(8);
  }
  __syncthreads();
  // phase B: my output base + grand total from the per-block counts
  if (tid == 0) {
    int basev = 0, total = 0;
    for (int j = 0; j < grid; ++j) {
      int v = __hip_atomic_load(&o.counts_ws[j], __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
      if (j < b) basev += v;
      total += v;
    }
    base_sh = basev;
    if (b == 0) *o.count_out = total;
  }
  __syncthreads();
  const int base_out = base_sh;
  // phase C: regenerate ALL fields for surviving rows; write the compacted
  // columns and the bf16 feature rows
  for (int j = tid; j < lcount; j += 256) {
    int64_t r = start + loc[j];
    int64_t dst = base_out + j;
    uint64_t rb = (c << 32) ^ (uint64_t)r;
    for (int f = 0; f < spec.nf; ++f) {
      uint64_t h = mix64(rb + ((uint64_t)(f + 1) << 52));
      float v = spec.lo[f] + spec.width[f] * u01(h);
      o.outs[f][dst] = v;
      if (o.feats) o.feats[dst * dpad + f] = (__bf16)v;
    }
    if (spec.key_range > 0) {
      uint64_t h = mix64(rb + (0x9E37ull << 48));
      o.key_out[dst] =
          spec.key_lo + (int64_t)(h % (uint64_t)spec.key_range);
    }
  }
  // advance the replay counter once per launch (grid ticket)
  __syncthreads();
  if (tid == 0) {
    unsigned long long t = atomicAdd(&o.ctr[1], 1ull);
    if (t == (unsigned long long)grid - 1) {
      o.ctr[1] = 0ull;
      o.ctr[0] = c + 1ull;
      __threadfence();
    }
  }
}

extern "C" int launch_genfiltpack(const float* lo, const float* width,
                                  int nf, int64_t key_lo, int64_t key_range,
                                  int64_t n, int fidx, int op, float scalar,
                                  float* const* outs, int64_t* key_out,
                                  void* feats, int dpad, int32_t* count_out,
                                  int32_t* counts_ws,
                                  unsigned long long* bar,
                                  unsigned long long* ctr, hipStream_t st) {
  int ch = GFP_CH;
  while (ch > 256 && (n + ch - 1) / ch < 64) ch >>= 1;
  int grid = (int)((n + ch - 1) / ch);
  // > 256 blocks: co-residency no longer trivially guaranteed for the
  // in-kernel barrier — caller falls back to the multi-kernel chain
  if (grid < 1 || grid > 256 || nf > 32) return -1;
  GenSpec spec{};
  spec.nf = nf;
  for (int f = 0; f < nf; ++f) {
    spec.lo[f] = lo[f];
    spec.width[f] = width[f];
  }
  spec.key_lo = key_lo;
  spec.key_range = key_range;
  GfpOuts o{};
  for (int f = 0; f < nf; ++f) o.outs[f] = outs[f];
  o.key_out = key_out;
  o.feats = (__bf16*)feats;
  o.count_out = count_out;
  o.counts_ws = counts_ws;
  o.bar = bar;
  o.ctr = ctr;
  genfiltpack_kernel<<<grid, 256, 0, st>>>(spec, n, ch, fidx, op, scalar,
                                           dpad, o);
  return 0;
}

// ---- featpack: nf gathered f32 columns → [n, kpad] bf16 --------------------
struct PackSpec {
  const float* src[32];
  int nf;
  int kpad;
};

__global__ void featpack_kernel(PackSpec spec, int64_t n,
                                __bf16* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    __bf16* row = out + i * spec.kpad;
#pragma unroll 4
    for (int f = 0; f < spec.nf; ++f) row[f] = (__bf16)spec.src[f][i];
    for (int f = spec.nf; f < spec.kpad; ++f) row[f] = (__bf16)0.f;
  }
}

extern "C" void launch_featpack(const float** src, int nf, int kpad,
                                int64_t n, void* out, hipStream_t st) {
  PackSpec spec{};
  spec.nf = nf > 32 ? 32 : nf;
  spec.kpad = kpad;
  for (int f = 0; f < spec.nf; ++f) spec.src[f] = src[f];
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  featpack_kernel<<<grid, 256, 0, st>>>(spec, n, (__bf16*)out);
}

// f64 source variant (protobuf floats decode to double)
struct PackSpec64 {
  const double* src[32];
  int nf;
  int kpad;
};

__global__ void featpack64_kernel(PackSpec64 spec, int64_t n,
                                  __bf16* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    __bf16* row = out + i * spec.kpad;
#pragma unroll 4
    for (int f = 0; f < spec.nf; ++f) row[f] = (__bf16)(float)spec.src[f][i];
    for (int f = spec.nf; f < spec.kpad; ++f) row[f] = (__bf16)0.f;
  }
}

extern "C" void launch_featpack64(const double** src, int nf, int kpad,
                                  int64_t n, void* out, hipStream_t st) {
  PackSpec64 spec{};
  spec.nf = nf > 32 ? 32 : nf;
  spec.kpad = kpad;
  for (int f = 0; f < spec.nf; ++f) spec.src[f] = src[f];
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  featpack64_kernel<<<grid, 256, 0, st>>>(spec, n, (__bf16*)out);
}

// ---- gemv: scores = x[M,K]·w[K] + b → f32 ----------------------------------
// one wave per 8 rows; lanes split K. K ≤ a few thousand (MLP hidden).
__global__ void gemv_bf16_f32_kernel(const __bf16* __restrict__ x,
                                     const __bf16* __restrict__ w,
                                     float bias, int64_t M, int K,
                                     float* __restrict__ out) {
  // wave w handles rows [w*8, w*8+8); each lane accumulates over K/8 slices
  int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  int64_t row0 = wave * 8;
  if (row0 >= M) return;
  int sub = lane >> 3;        // 8 lanes per row
  int piece = lane & 7;
  int64_t row = row0 + sub;
  if (row >= M) return;
  const __bf16* xr = x + row * K;
  float acc = 0.f;
  for (int k = piece; k < K; k += 8) acc += (float)xr[k] * (float)w[k];
  // reduce across the 8 consecutive lanes owning this row
  for (int off = 1; off < 8; off <<= 1) {
    float o = __shfl_down(acc, off, 64);
    if (piece + off < 8) acc += o;
  }
  if (piece == 0) out[row] = acc + bias;
}

extern "C" void launch_gemv_bf16_f32(const void* x, const void* w, float bias,
                                     int64_t M, int K, float* out,
                                     hipStream_t st) {
  int64_t waves = (M + 7) / 8;
  int64_t threads = waves * 64;
  int grid = (int)((threads + 255) / 256);
  if (grid < 1) return;
  gemv_bf16_f32_kernel<<<grid, 256, 0, st>>>((const __bf16*)x,
                                             (const __bf16*)w, bias, M, K,
                                             out);
}
