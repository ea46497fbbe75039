// Device hash join (build/probe) — replaces DataFusion's HashJoinExec for the
// sql processor's JOINs and the window join buffers (reference
// buffer/join.rs:62-130, processor/sql.rs:148-183 temporary joins).
//
// Build side = right table: open-addressing table of key→chain-head plus a
// per-row `next` array (chaining handles duplicate keys). Probe = two passes
// (count → scan by caller → emit) so output pairs are dense and left-ordered.
#include "common.h"

#define EMPTY_KEY 0x8000000000000000ll
#define JOIN_BLOCK 256

__global__ void join_build_kernel(const int64_t* __restrict__ keys, int64_t n,
                                  int64_t* __restrict__ table_keys,
                                  int32_t* __restrict__ table_head,
                                  int32_t* __restrict__ next,
                                  uint32_t table_mask) {
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
        if (prev == EMPTY_KEY || prev == k) break;
      }
      h = (h + 1) & table_mask;
    }
    // push row i onto the chain at slot h
    int32_t old = atomicExch(&table_head[h], (int32_t)i);
    next[i] = old;
  }
}

__global__ void join_probe_count_kernel(const int64_t* __restrict__ lkeys,
                                        int64_t n_left,
                                        const int64_t* __restrict__ table_keys,
                                        const int32_t* __restrict__ table_head,
                                        const int32_t* __restrict__ next,
                                        const int64_t* __restrict__ rkeys,
                                        uint32_t table_mask,
                                        int32_t* __restrict__ counts) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_left; i += stride) {
    int64_t k = lkeys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    int cnt = 0;
    for (;;) {
      int64_t cur = table_keys[h];
      if (cur == EMPTY_KEY) break;
      if (cur == k) {
        for (int32_t r = table_head[h]; r >= 0; r = next[r])
          if (rkeys[r] == k) ++cnt;
        break;
      }
      h = (h + 1) & table_mask;
    }
    counts[i] = cnt;
  }
}

__global__ void join_probe_emit_kernel(const int64_t* __restrict__ lkeys,
                                       int64_t n_left,
                                       const int64_t* __restrict__ table_keys,
                                       const int32_t* __restrict__ table_head,
                                       const int32_t* __restrict__ next,
                                       const int64_t* __restrict__ rkeys,
                                       uint32_t table_mask,
                                       const int32_t* __restrict__ offsets,
                                       int64_t* __restrict__ l_out,
                                       int64_t* __restrict__ r_out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_left; i += stride) {
    int64_t k = lkeys[i];
    uint32_t h = (uint32_t)(mix64((uint64_t)k) & table_mask);
    int32_t w = offsets[i];
    for (;;) {
      int64_t cur = table_keys[h];
      if (cur == EMPTY_KEY) break;
      if (cur == k) {
        for (int32_t r = table_head[h]; r >= 0; r = next[r])
          if (rkeys[r] == k) {
            l_out[w] = i;
            r_out[w] = r;
            ++w;
          }
        break;
      }
      h = (h + 1) & table_mask;
    }
  }
}

__global__ void join_fill_kernel(int64_t* p, int64_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;  // grid-stride: grid capped at 2048
}
__global__ void join_fill32_kernel(int32_t* p, int32_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;
}

extern "C" {

static int jgrid(int64_t n) {
  int64_t g = (n + JOIN_BLOCK - 1) / JOIN_BLOCK;
  return (int)(g > 2048 ? 2048 : (g < 1 ? 1 : g));
}

void launch_join_build(const int64_t* rkeys, int64_t n_right,
                       int64_t* table_keys, int32_t* table_head, int32_t* next,
                       uint32_t table_size, hipStream_t st) {
  join_fill_kernel<<<jgrid(table_size), JOIN_BLOCK, 0, st>>>(
      table_keys, EMPTY_KEY, table_size);
  join_fill32_kernel<<<jgrid(table_size), JOIN_BLOCK, 0, st>>>(
      table_head, -1, table_size);
  join_build_kernel<<<jgrid(n_right), JOIN_BLOCK, 0, st>>>(
      rkeys, n_right, table_keys, table_head, next, table_size - 1);
}

void launch_join_probe_count(const int64_t* lkeys, int64_t n_left,
                             const int64_t* table_keys,
                             const int32_t* table_head, const int32_t* next,
                             const int64_t* rkeys, uint32_t table_size,
                             int32_t* counts, hipStream_t st) {
  join_probe_count_kernel<<<jgrid(n_left), JOIN_BLOCK, 0, st>>>(
      lkeys, n_left, table_keys, table_head, next, rkeys, table_size - 1,
      counts);
}

void launch_join_probe_emit(const int64_t* lkeys, int64_t n_left,
                            const int64_t* table_keys,
                            const int32_t* table_head, const int32_t* next,
                            const int64_t* rkeys, uint32_t table_size,
                            const int32_t* offsets, int64_t* l_out,
                            int64_t* r_out, hipStream_t st) {
  join_probe_emit_kernel<<<jgrid(n_left), JOIN_BLOCK, 0, st>>>(
      lkeys, n_left, table_keys, table_head, next, rkeys, table_size - 1,
      offsets, l_out, r_out);
}

}  // extern "C"
