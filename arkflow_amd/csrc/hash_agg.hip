// GPU hash GROUP BY: open-addressing device hash table + segment reductions.
//
// Replaces DataFusion's hash-aggregate physical operator on the `sql`
// processor path (reference processor/sql.rs execute_query → AggregateExec).
// Build: per-row atomicCAS claim into a power-of-two table; first claimant
// assigns a dense group id. Reduce: LDS-tiled per-block partials when the
// group count is small (the common GROUP BY case), global atomics otherwise.
#include "common.h"

#define EMPTY_KEY 0x8000000000000000ll
#define AGG_BLOCK 256
#define LDS_GROUPS 2048

enum RedOp { SUM = 0, MIN = 1, MAX = 2 };

static int32_t float_flip_host(float f) {
  union { float f; uint32_t u; } x{f};
  uint32_t v = (x.u & 0x80000000u) ? ~x.u : (x.u | 0x80000000u);
  return (int32_t)v;  // bit pattern; device compares as unsigned
}

// ---- group-id assignment ------------------------------------------------
// Two kernels: INSERT claims keys and assigns dense gids; LOOKUP probes with
// plain loads after the kernel boundary (which orders all insert writes) —
// no per-row acquire/invalidate in the hot path.
__global__ void hash_insert_kernel(const int64_t* __restrict__ keys, int64_t n,
                                   int64_t* __restrict__ table_keys,
                                   int32_t* __restrict__ table_gids,
                                   uint32_t table_mask,
                                   int32_t* __restrict__ counter) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    for (;;) {
      int64_t cur = __hip_atomic_load(&table_keys[h], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT);
      if (cur == k) break;
      if (cur == EMPTY_KEY) {
        int64_t prev = (int64_t)atomicCAS(
            (unsigned long long*)&table_keys[h], (unsigned long long)EMPTY_KEY,
            (unsigned long long)k);
        if (prev == EMPTY_KEY) {
          int32_t gid = atomicAdd(counter, 1);
          __hip_atomic_store(&table_gids[h], gid, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
        if (prev == k) break;
      }
      h = (h + 1) & table_mask;
    }
  }
}

__global__ void hash_lookup_kernel(const int64_t* __restrict__ keys, int64_t n,
                                   const int64_t* __restrict__ table_keys,
                                   const int32_t* __restrict__ table_gids,
                                   uint32_t table_mask,
                                   int32_t* __restrict__ gids_out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    while (table_keys[h] != k) h = (h + 1) & table_mask;
    gids_out[i] = table_gids[h];
  }
}

// uniq[gid] = key
__global__ void hash_export_kernel(const int64_t* __restrict__ table_keys,
                                   const int32_t* __restrict__ table_gids,
                                   uint32_t table_size,
                                   int64_t* __restrict__ uniq) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t stride = gridDim.x * blockDim.x;
  for (; i < table_size; i += stride) {
    int64_t k = table_keys[i];
    if (k != EMPTY_KEY) uniq[table_gids[i]] = k;
  }
}

// ---- segment reductions ------------------------------------------------------
// SUM, LDS-tiled: per-block float partials, one global atomic per group/block.
__global__ void segment_sum_lds_kernel(const float* __restrict__ vals,
                                       const int32_t* __restrict__ gids,
                                       int64_t n, int g,
                                       float* __restrict__ out) {
  __shared__ float part[LDS_GROUPS];
  for (int j = threadIdx.x; j < g; j += blockDim.x) part[j] = 0.f;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) atomicAdd(&part[gids[i]], vals[i]);
  __syncthreads();
  for (int j = threadIdx.x; j < g; j += blockDim.x)
    if (part[j] != 0.f) atomicAdd(&out[j], part[j]);
}

__global__ void segment_sum_global_kernel(const float* __restrict__ vals,
                                          const int32_t* __restrict__ gids,
                                          int64_t n, float* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) atomicAdd(&out[gids[i]], vals[i]);
}

// MIN/MAX on monotone-flipped int32 (handles negative floats correctly).
__global__ void segment_mm_lds_kernel(const float* __restrict__ vals,
                                      const int32_t* __restrict__ gids,
                                      int64_t n, int g, int op,
                                      int32_t* __restrict__ out_flipped,
                                      int32_t init) {
  __shared__ int32_t part[LDS_GROUPS];
  for (int j = threadIdx.x; j < g; j += blockDim.x) part[j] = init;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint32_t v = float_flip(vals[i]);
    if (op == MIN) atomicMin((uint32_t*)&part[gids[i]], v);
    else atomicMax((uint32_t*)&part[gids[i]], v);
  }
  __syncthreads();
  for (int j = threadIdx.x; j < g; j += blockDim.x) {
    if (op == MIN) atomicMin((uint32_t*)&out_flipped[j], (uint32_t)part[j]);
    else atomicMax((uint32_t*)&out_flipped[j], (uint32_t)part[j]);
  }
}

__global__ void segment_mm_global_kernel(const float* __restrict__ vals,
                                         const int32_t* __restrict__ gids,
                                         int64_t n, int op,
                                         int32_t* __restrict__ out_flipped) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint32_t v = float_flip(vals[i]);
    if (op == MIN) atomicMin((uint32_t*)&out_flipped[gids[i]], v);
    else atomicMax((uint32_t*)&out_flipped[gids[i]], v);
  }
}

__global__ void unflip_kernel(const int32_t* __restrict__ in,
                              float* __restrict__ out, int g) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < g) out[i] = float_unflip((uint32_t)in[i]);
}

__global__ void fill_i64_kernel(int64_t* p, int64_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;  // grid-stride: grid is capped at 2048
}
__global__ void fill_i32_kernel(int32_t* p, int32_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;
}

__global__ void fill_f32_kernel(float* p, float v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;
}

// ---- capture-safe (device-count) variants ------------------------------------
// Same algorithms with the row count read from a device pointer (the
// capture-safe filter's surviving-row count), so the whole
// filter -> group-by -> reduce chain records into one hipGraph with static
// grids. Garbage rows past *nrow in the padded inputs are never touched.
__global__ void hash_insert_dev_kernel(const int64_t* __restrict__ keys,
                                       const int32_t* __restrict__ nrow,
                                       int64_t* __restrict__ table_keys,
                                       int32_t* __restrict__ table_gids,
                                       uint32_t table_mask,
                                       int32_t* __restrict__ counter) {
  const int64_t n = *nrow;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    for (;;) {
      int64_t cur = __hip_atomic_load(&table_keys[h], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT);
      if (cur == k) break;
      if (cur == EMPTY_KEY) {
        int64_t prev = (int64_t)atomicCAS(
            (unsigned long long*)&table_keys[h], (unsigned long long)EMPTY_KEY,
            (unsigned long long)k);
        if (prev == EMPTY_KEY) {
          int32_t gid = atomicAdd(counter, 1);
          __hip_atomic_store(&table_gids[h], gid, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
        if (prev == k) break;
      }
      h = (h + 1) & table_mask;
    }
  }
}

__global__ void hash_lookup_dev_kernel(const int64_t* __restrict__ keys,
                                       const int32_t* __restrict__ nrow,
                                       const int64_t* __restrict__ table_keys,
                                       const int32_t* __restrict__ table_gids,
                                       uint32_t table_mask,
                                       int32_t* __restrict__ gids_out) {
  const int64_t n = *nrow;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t k = keys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    while (table_keys[h] != k) h = (h + 1) & table_mask;
    gids_out[i] = table_gids[h];
  }
}

// count + sum in one LDS-tiled pass (vals may be null: count only)
__global__ void segment_cs_lds_dev_kernel(const float* __restrict__ vals,
                                          const int32_t* __restrict__ gids,
                                          const int32_t* __restrict__ nrow,
                                          int g_cap,
                                          float* __restrict__ counts,
                                          float* __restrict__ sums) {
  __shared__ float pc[LDS_GROUPS];
  __shared__ float ps[LDS_GROUPS];
  for (int j = threadIdx.x; j < g_cap; j += blockDim.x) {
    pc[j] = 0.f;
    ps[j] = 0.f;
  }
  __syncthreads();
  const int64_t n = *nrow;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int g = gids[i];
    if ((unsigned)g >= (unsigned)g_cap) continue;  // overflow guard; the
    // Python wrapper validates g <= g_cap after every step
    atomicAdd(&pc[g], 1.f);
    if (vals) atomicAdd(&ps[g], vals[i]);
  }
  __syncthreads();
  for (int j = threadIdx.x; j < g_cap; j += blockDim.x) {
    if (pc[j] != 0.f) {
      if (counts) atomicAdd(&counts[j], pc[j]);
      if (sums && vals) atomicAdd(&sums[j], ps[j]);
    }
  }
}

__global__ void segment_sum_lds_dev_kernel(const float* __restrict__ vals,
                                           const int32_t* __restrict__ gids,
                                           const int32_t* __restrict__ nrow,
                                           int g_cap, float* __restrict__ out) {
  __shared__ float part[LDS_GROUPS];
  for (int j = threadIdx.x; j < g_cap; j += blockDim.x) part[j] = 0.f;
  __syncthreads();
  const int64_t n = *nrow;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    if ((unsigned)gids[i] < (unsigned)g_cap)
      atomicAdd(&part[gids[i]], vals[i]);
  __syncthreads();
  for (int j = threadIdx.x; j < g_cap; j += blockDim.x)
    if (part[j] != 0.f) atomicAdd(&out[j], part[j]);
}

__global__ void segment_mm_global_dev_kernel(
    const float* __restrict__ vals, const int32_t* __restrict__ gids,
    const int32_t* __restrict__ nrow, int op, int g_cap,
    int32_t* __restrict__ out_flipped) {
  const int64_t n = *nrow;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int g = gids[i];
    if ((unsigned)g >= (unsigned)g_cap) continue;
    uint32_t v = float_flip(vals[i]);
    if (op == MIN) atomicMin((uint32_t*)&out_flipped[g], v);
    else atomicMax((uint32_t*)&out_flipped[g], v);
  }
}

// export with a bound (capture path: uniq buffer is g_cap entries)
__global__ void hash_export_cap_kernel(const int64_t* __restrict__ table_keys,
                                       const int32_t* __restrict__ table_gids,
                                       uint32_t table_size, int g_cap,
                                       int64_t* __restrict__ uniq) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t stride = gridDim.x * blockDim.x;
  for (; i < table_size; i += stride) {
    int64_t k = table_keys[i];
    int32_t g = k != EMPTY_KEY ? table_gids[i] : -1;
    if (g >= 0 && g < g_cap) uniq[g] = k;
  }
}

// one-kernel reset of every per-step buffer (table, counter, counts, up to
// 4 reduction buffers) — replaces 5-7 fill launches; at ~100-group tables
// each extra launch is pure overhead
__global__ void agg_reset_kernel(int64_t* __restrict__ tk,
                                 int32_t* __restrict__ tg, uint32_t ts,
                                 int32_t* __restrict__ counter,
                                 float* __restrict__ counts, int g_cap,
                                 float* s0, float* s1, float* s2, float* s3,
                                 int32_t* m0, int32_t* m1, int32_t* m2,
                                 int32_t* m3, int32_t i0, int32_t i1,
                                 int32_t i2, int32_t i3) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  if (i == 0) *counter = 0;
  for (int64_t j = i; j < ts; j += stride) {
    tk[j] = EMPTY_KEY;
    tg[j] = -1;
  }
  for (int64_t j = i; j < g_cap; j += stride) {
    counts[j] = 0.f;
    if (s0) s0[j] = 0.f;
    if (s1) s1[j] = 0.f;
    if (s2) s2[j] = 0.f;
    if (s3) s3[j] = 0.f;
    if (m0) m0[j] = i0;
    if (m1) m1[j] = i1;
    if (m2) m2[j] = i2;
    if (m3) m3[j] = i3;
  }
}

// SQL count() is int64: convert in-graph so no per-step torch cast runs
__global__ void counts_to_i64_kernel(const float* __restrict__ counts,
                                     int g_cap, int64_t* __restrict__ out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < g_cap) out[i] = (int64_t)(counts[i] + 0.5f);
}

__global__ void copy_i32_kernel(const int32_t* src, int32_t* dst) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *dst = *src;
}

// ---- host launchers ----------------------------------------------------------
extern "C" {

static int grid_for(int64_t n, int block) {
  int64_t g = (n + block - 1) / block;
  return (int)(g > 2048 ? 2048 : (g < 1 ? 1 : g));
}

void launch_hash_build(const int64_t* keys, int64_t n, int64_t* table_keys,
                       int32_t* table_gids, uint32_t table_size,
                       int32_t* counter, int32_t* gids_out, hipStream_t st) {
  fill_i64_kernel<<<grid_for(table_size, 256), 256, 0, st>>>(
      table_keys, EMPTY_KEY, table_size);
  fill_i32_kernel<<<grid_for(table_size, 256), 256, 0, st>>>(
      table_gids, -1, table_size);
  hash_insert_kernel<<<grid_for(n, AGG_BLOCK), AGG_BLOCK, 0, st>>>(
      keys, n, table_keys, table_gids, table_size - 1, counter);
  hash_lookup_kernel<<<grid_for(n, AGG_BLOCK), AGG_BLOCK, 0, st>>>(
      keys, n, table_keys, table_gids, table_size - 1, gids_out);
}

void launch_hash_export(const int64_t* table_keys, const int32_t* table_gids,
                        uint32_t table_size, int64_t* uniq, hipStream_t st) {
  hash_export_kernel<<<grid_for(table_size, 256), 256, 0, st>>>(
      table_keys, table_gids, table_size, uniq);
}

// op: 0=sum 1=min 2=max; out must be zero-filled for sum.
// scratch_flipped: int32[g] workspace for min/max.
void launch_segment_reduce_f32(const float* vals, const int32_t* gids,
                               int64_t n, int g, int op, float* out,
                               int32_t* scratch_flipped, hipStream_t st) {
  int grid = grid_for(n, AGG_BLOCK);
  if (op == SUM) {
    if (g <= LDS_GROUPS)
      segment_sum_lds_kernel<<<grid, AGG_BLOCK, 0, st>>>(vals, gids, n, g, out);
    else
      segment_sum_global_kernel<<<grid, AGG_BLOCK, 0, st>>>(vals, gids, n, out);
    return;
  }
  int32_t init = float_flip_host(op == MIN ? INFINITY : -INFINITY);
  fill_i32_kernel<<<grid_for(g, 256), 256, 0, st>>>(scratch_flipped, init, g);
  if (g <= LDS_GROUPS)
    segment_mm_lds_kernel<<<grid, AGG_BLOCK, 0, st>>>(
        vals, gids, n, g, op, scratch_flipped, init);
  else
    segment_mm_global_kernel<<<grid, AGG_BLOCK, 0, st>>>(
        vals, gids, n, op, scratch_flipped);
  unflip_kernel<<<grid_for(g, 256), 256, 0, st>>>(scratch_flipped, out, g);
}

// Capture-safe whole-chain launcher: reset (static fills) -> insert/lookup
// (device n) -> count+sum fused -> extra reductions -> export -> counter copy.
// ops[i]: 0=sum 1=min 2=max. red_out[i] is the f32[g_cap] destination for
// vals[i]; mm_scratch[i] only used (non-null) for min/max.
void launch_hash_agg_capture(const int64_t* keys, const int32_t* nrow,
                             int64_t n_cap, int64_t* table_keys,
                             int32_t* table_gids, uint32_t table_size,
                             int32_t* counter, int32_t* gids, float* counts,
                             int g_cap, const float* const* vals,
                             const int* ops, float* const* red_out,
                             int32_t* const* mm_scratch, int nv,
                             int64_t* uniq, int64_t* counts_i64,
                             int32_t* gcount_out, hipStream_t st) {
  // one fused reset launch for every per-step buffer (nv <= 4)
  float* sb[4] = {nullptr, nullptr, nullptr, nullptr};
  int32_t* mb[4] = {nullptr, nullptr, nullptr, nullptr};
  int32_t mi[4] = {0, 0, 0, 0};
  for (int i = 0; i < nv && i < 4; ++i) {
    if (ops[i] == SUM) {
      sb[i] = red_out[i];
    } else {
      mb[i] = mm_scratch[i];
      mi[i] = float_flip_host(ops[i] == MIN ? INFINITY : -INFINITY);
    }
  }
  agg_reset_kernel<<<grid_for(table_size > g_cap ? table_size : g_cap, 256),
                     256, 0, st>>>(table_keys, table_gids, table_size,
                                   counter, counts, g_cap, sb[0], sb[1],
                                   sb[2], sb[3], mb[0], mb[1], mb[2], mb[3],
                                   mi[0], mi[1], mi[2], mi[3]);
  int grid = grid_for(n_cap, AGG_BLOCK);
  hash_insert_dev_kernel<<<grid, AGG_BLOCK, 0, st>>>(
      keys, nrow, table_keys, table_gids, table_size - 1, counter);
  hash_lookup_dev_kernel<<<grid, AGG_BLOCK, 0, st>>>(
      keys, nrow, table_keys, table_gids, table_size - 1, gids);
  // first sum column rides the count pass
  int first_sum = -1;
  for (int i = 0; i < nv && first_sum < 0; ++i)
    if (ops[i] == SUM) first_sum = i;
  segment_cs_lds_dev_kernel<<<grid, AGG_BLOCK, 0, st>>>(
      first_sum >= 0 ? vals[first_sum] : nullptr, gids, nrow, g_cap, counts,
      first_sum >= 0 ? red_out[first_sum] : nullptr);
  for (int i = 0; i < nv; ++i) {
    if (i == first_sum) continue;
    if (ops[i] == SUM) {
      segment_sum_lds_dev_kernel<<<grid, AGG_BLOCK, 0, st>>>(
          vals[i], gids, nrow, g_cap, red_out[i]);
    } else {
      segment_mm_global_dev_kernel<<<grid, AGG_BLOCK, 0, st>>>(
          vals[i], gids, nrow, (int)ops[i], g_cap, mm_scratch[i]);
      unflip_kernel<<<grid_for(g_cap, 256), 256, 0, st>>>(mm_scratch[i],
                                                          red_out[i], g_cap);
    }
  }
  hash_export_cap_kernel<<<grid_for(table_size, 256), 256, 0, st>>>(
      table_keys, table_gids, table_size, g_cap, uniq);
  if (counts_i64)
    counts_to_i64_kernel<<<grid_for(g_cap, 256), 256, 0, st>>>(counts, g_cap,
                                                               counts_i64);
  if (gcount_out)
    copy_i32_kernel<<<1, 1, 0, st>>>(counter, gcount_out);
}

}  // extern "C"
