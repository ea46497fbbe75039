// Torch extension bindings: arkflow_amd._native.
// Thin host layer over the gfx950 kernels in *.hip — tensor checks, workspace
// allocation, the tiny device cumsum between two-pass kernels, pybind exports.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>
#include <tuple>
#include <vector>

// ---- extern kernel launchers (defined in the .hip TUs) ----------------------
extern "C" {
int filter_grid(int64_t n);
void launch_filter_count_f32(const float*, int64_t, int, float, int32_t*,
                             hipStream_t);
void launch_filter_scatter_f32(const float*, int64_t, int, float,
                               const int32_t*, int32_t*, hipStream_t);
void launch_filter_count_i64(const int64_t*, int64_t, int, int64_t, int32_t*,
                             hipStream_t);
void launch_filter_scatter_i64(const int64_t*, int64_t, int, int64_t,
                               const int32_t*, int32_t*, hipStream_t);
void launch_mask_count(const bool*, int64_t, int32_t*, hipStream_t);
void launch_mask_scatter(const bool*, int64_t, const int32_t*, int32_t*,
                         hipStream_t);
void launch_gather(const void*, const int32_t*, int64_t, void*, int,
                   hipStream_t);
void launch_hash_build(const int64_t*, int64_t, int64_t*, int32_t*, uint32_t,
                       int32_t*, int32_t*, hipStream_t);
void launch_hash_export(const int64_t*, const int32_t*, uint32_t, int64_t*,
                        hipStream_t);
void launch_segment_reduce_f32(const float*, const int32_t*, int64_t, int, int,
                               float*, int32_t*, hipStream_t);
void launch_hash_agg_capture(const int64_t*, const int32_t*, int64_t,
                             int64_t*, int32_t*, uint32_t, int32_t*, int32_t*,
                             float*, int, const float* const*, const int*,
                             float* const*, int32_t* const*, int, int64_t*,
                             int64_t*, int32_t*, hipStream_t);
void launch_join_build(const int64_t*, int64_t, int64_t*, int32_t*, int32_t*,
                       uint32_t, hipStream_t);
void launch_join_probe_count(const int64_t*, int64_t, const int64_t*,
                             const int32_t*, const int32_t*, const int64_t*,
                             uint32_t, int32_t*, hipStream_t);
void launch_join_probe_emit(const int64_t*, int64_t, const int64_t*,
                            const int32_t*, const int32_t*, const int64_t*,
                            uint32_t, const int32_t*, int64_t*, int64_t*,
                            hipStream_t);
void launch_gemm_bf16(const void*, const void*, const float*, void*, int, int,
                      int, int, hipStream_t);
int launch_gemm_bf16_8p(const void*, const void*, const float*, void*, int,
                        int, int, int, int, hipStream_t);
int launch_gemm_bf16_2p(const void*, const void*, const float*, void*, int,
                        int, int, int, hipStream_t);
void launch_layernorm_bf16(const void*, const void*, const float*,
                           const float*, void*, void*, int64_t, int, float,
                           hipStream_t);
void launch_softmax_bf16(const void*, void*, int64_t, int, float, hipStream_t);
void launch_bias_act_bf16(const void*, const float*, void*, int64_t, int, int,
                          hipStream_t);
int launch_attention_bf16(const void*, const void*, const void*, void*, int,
                          int, int, float, hipStream_t);
int launch_attention_qkv_bf16(const void*, void*, int, int, int, int, float,
                              hipStream_t);
int launch_attention_qkv_bf16_pad(const void*, void*, int, int, int, int,
                                  float, int, hipStream_t);
void launch_proto_copy_bytes(const uint8_t*, const int64_t*, const int64_t*,
                             int64_t, uint8_t*, hipStream_t);
void launch_proto_decode(const uint8_t*, const int64_t*, int64_t, int,
                         const int*, const int*, const int*, const int*,
                         int64_t*, double*, int64_t*, int32_t*, int32_t*,
                         hipStream_t);
void launch_gather_multi(int, const void**, void**, const int*,
                         const int32_t*, int64_t, hipStream_t);
void launch_gather_multi_dyn(int, const void**, void**, const int*,
                             const int32_t*, const int32_t*, int64_t,
                             hipStream_t);
void launch_gemm_bf16_skinny(const void*, const void*, const float*, void*,
                             int, int, int, int, hipStream_t);
int launch_gemm_bf16_k64(const void*, const void*, const float*, void*, int,
                         int, int, int, hipStream_t);
int launch_gemm_bf16_k64s(const void*, const void*, const float*, void*,
                          int, int, int, int, hipStream_t);
int launch_gemm_bf16_k64p(const void*, const void*, const float*, void*,
                          int, int, int, int, hipStream_t);
int launch_gemm_bf16_k64w(const void*, const void*, const float*, void*,
                          int, int, int, int, hipStream_t);
int launch_gemm_bf16_k64s3(const void*, const void*, const float*, void*,
                          int, int, int, int, hipStream_t);
int launch_gemm_bf16_k64d(const void*, const void*, const float*, void*,
                          int, int, int, int, hipStream_t);
void launch_gen_fields(float*, int64_t*, int64_t, const float*, const float*,
                       int, int64_t, int64_t, unsigned long long*,
                       hipStream_t);
int launch_genfiltpack(const float*, const float*, int, int64_t, int64_t,
                       int64_t, int, int, float, float* const*, int64_t*,
                       void*, int, int32_t*, int32_t*, unsigned long long*,
                       unsigned long long*, hipStream_t);
void launch_scan_counts(const int32_t*, int, int32_t*, int32_t*,
                        hipStream_t);
void launch_featpack(const float**, int, int, int64_t, void*, hipStream_t);
void launch_featpack64(const double**, int, int, int64_t, void*,
                       hipStream_t);
void launch_gemv_bf16_f32(const void*, const void*, float, int64_t, int,
                          float*, hipStream_t);
void launch_bytes_hash(const uint8_t*, const int64_t*, int64_t, int64_t*,
                       hipStream_t);
void launch_bytes_match(const uint8_t*, const int64_t*, int64_t,
                        const uint8_t*, int, int, bool*, hipStream_t);
int scan_grid(int64_t n);
void launch_scan_partials(const int32_t*, int64_t, int64_t*, hipStream_t);
void launch_scan_boffs64(const int64_t*, int, int64_t*, int64_t*,
                         hipStream_t);
void launch_scan_write(const int32_t*, int64_t, const int64_t*, int64_t*,
                       hipStream_t);
int radix_sort_nblocks(int64_t);
int onesweep_nblocks(int64_t);
void launch_os_hist256_u32(const uint32_t*, int64_t, int, int32_t*, int,
                           hipStream_t);
void launch_os_hist256_u64(const uint64_t*, int64_t, int, int32_t*, int,
                           hipStream_t);
void launch_onesweep_pass_u32(const uint32_t*, const int32_t*, uint32_t*,
                              int32_t*, int64_t, int, const int32_t*, int,
                              hipStream_t);
void launch_onesweep_pass_u64(const uint64_t*, const int32_t*, uint64_t*,
                              int32_t*, int64_t, int, const int32_t*, int,
                              hipStream_t);
void launch_i32_to_ordered(const int32_t*, uint32_t*, int64_t, int,
                           hipStream_t);
void launch_radix_hist_u32(const uint32_t*, const int32_t*, int64_t, int,
                           int32_t*, int, hipStream_t);
void launch_radix_scatter_u32(const uint32_t*, const int32_t*, int64_t, int,
                              const int32_t*, int, int32_t*, hipStream_t);
void launch_radix_hist_u64(const uint64_t*, const int32_t*, int64_t, int,
                           int32_t*, int, hipStream_t);
void launch_radix_scatter_u64(const uint64_t*, const int32_t*, int64_t, int,
                              const int32_t*, int, int32_t*, hipStream_t);
void launch_f32_to_ordered(const float*, uint32_t*, int64_t, int,
                           hipStream_t);
void launch_i64_to_ordered(const int64_t*, uint64_t*, int64_t, int,
                           hipStream_t);
void launch_json_decode(const uint8_t*, const int64_t*, int64_t, int,
                        const char*, const int*, const int*, const int*,
                        const int*, int, const char*, const int*,
                        double*, int64_t*, int64_t*, int32_t*, uint8_t*,
                        int32_t*, int32_t*, int, hipStream_t);
void launch_json_copy_strings(const uint8_t*, const int64_t*, const int64_t*,
                              const uint8_t*, int64_t, int64_t, uint8_t*,
                              hipStream_t);
void launch_json_copy_strings_wave(const uint8_t*, const int64_t*,
                                   const int64_t*, const uint8_t*, int64_t,
                                   int64_t, uint8_t*, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_cuda(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

uint32_t next_pow2(uint32_t v) {
  v--;
  v |= v >> 1; v |= v >> 2; v |= v >> 4; v |= v >> 8; v |= v >> 16;
  return v + 1;
}

// exclusive prefix-sum of int32 lengths → int64 offsets [n+1] (offsets[n] =
// total), all on device — csrc/filter.hip scan kernels (torch's innermost
// cumsum is ~40x slower at n~262K)
torch::Tensor exclusive_offsets(torch::Tensor lens) {
  int64_t n = lens.numel();
  auto out = torch::empty({n + 1}, lens.options().dtype(torch::kInt64));
  if (n == 0) return out.zero_();
  auto st = cur_stream();
  int grid = scan_grid(n);
  auto partials = torch::empty({grid}, lens.options().dtype(torch::kInt64));
  launch_scan_partials(lens.data_ptr<int32_t>(), n,
                       partials.data_ptr<int64_t>(), st);
  torch::Tensor boffs;
  if (grid <= 1024) {
    // single-block scan writes offsets AND the total slot in one launch
    boffs = torch::empty({grid}, partials.options());
    launch_scan_boffs64(partials.data_ptr<int64_t>(), grid,
                        boffs.data_ptr<int64_t>(),
                        out.data_ptr<int64_t>() + n, st);
  } else {
    auto inc = partials.cumsum(0);
    boffs = inc - partials;
    out.narrow(0, n, 1).copy_(inc.narrow(0, grid - 1, 1));
  }
  launch_scan_write(lens.data_ptr<int32_t>(), n, boffs.data_ptr<int64_t>(),
                    out.data_ptr<int64_t>(), st);
  return out;
}

// exclusive scan of int32 block counts → (offsets, total tensor on device)
std::tuple<torch::Tensor, torch::Tensor> exscan(const torch::Tensor& counts) {
  auto inc = counts.cumsum(0, torch::kInt32);
  auto offs = inc - counts;
  auto total = inc.numel() > 0
                   ? inc.slice(0, inc.numel() - 1, inc.numel())
                   : torch::zeros({1}, counts.options());
  return {offs, total};
}

// ---------------------------------------------------------------------- filter
torch::Tensor mask_to_indices(torch::Tensor mask) {
  check_cuda(mask, "mask");
  TORCH_CHECK(mask.scalar_type() == torch::kBool, "mask must be bool");
  int64_t n = mask.numel();
  auto st = cur_stream();
  int nblocks = filter_grid(n);
  auto counts = torch::empty({std::max(nblocks, 1)},
                             mask.options().dtype(torch::kInt32));
  if (n == 0) return torch::empty({0}, mask.options().dtype(torch::kInt32));
  launch_mask_count(mask.data_ptr<bool>(), n, counts.data_ptr<int32_t>(), st);
  auto [offs, total_t] = exscan(counts);
  int64_t total = total_t.item<int32_t>();  // one device sync per filter op
  auto out = torch::empty({total}, mask.options().dtype(torch::kInt32));
  if (total)
    launch_mask_scatter(mask.data_ptr<bool>(), n, offs.data_ptr<int32_t>(),
                        out.data_ptr<int32_t>(), st);
  return out;
}

torch::Tensor filter_cmp_scalar(torch::Tensor col, int64_t op, double scalar) {
  check_cuda(col, "col");
  int64_t n = col.numel();
  if (n == 0) return torch::empty({0}, col.options().dtype(torch::kInt32));
  auto st = cur_stream();
  int nblocks = filter_grid(n);
  auto counts = torch::empty({nblocks}, col.options().dtype(torch::kInt32));
  if (col.scalar_type() == torch::kInt32) col = col.to(torch::kInt64);
  if (col.scalar_type() == torch::kFloat32) {
    launch_filter_count_f32(col.data_ptr<float>(), n, (int)op, (float)scalar,
                            counts.data_ptr<int32_t>(), st);
  } else if (col.scalar_type() == torch::kInt64) {
    launch_filter_count_i64(col.data_ptr<int64_t>(), n, (int)op,
                            (int64_t)scalar, counts.data_ptr<int32_t>(), st);
  } else {
    TORCH_CHECK(false, "filter_cmp_scalar supports f32/i64/i32 cols");
  }
  auto [offs, total_t] = exscan(counts);
  int64_t total = total_t.item<int32_t>();
  auto out = torch::empty({total}, col.options().dtype(torch::kInt32));
  if (total) {
    if (col.scalar_type() == torch::kFloat32)
      launch_filter_scatter_f32(col.data_ptr<float>(), n, (int)op,
                                (float)scalar, offs.data_ptr<int32_t>(),
                                out.data_ptr<int32_t>(), st);
    else
      launch_filter_scatter_i64(col.data_ptr<int64_t>(), n, (int)op,
                                (int64_t)scalar, offs.data_ptr<int32_t>(),
                                out.data_ptr<int32_t>(), st);
  }
  return out;
}

torch::Tensor gather(torch::Tensor col, torch::Tensor idx) {
  check_cuda(col, "col");
  check_cuda(idx, "idx");
  auto idx32 = idx.scalar_type() == torch::kInt32 ? idx
                                                  : idx.to(torch::kInt32);
  int64_t m = idx32.numel();
  auto out = torch::empty({m}, col.options());
  int esz = (int)col.element_size();
  TORCH_CHECK(esz == 1 || esz == 2 || esz == 4 || esz == 8,
              "unsupported element size");
  if (m)
    launch_gather(col.data_ptr(), idx32.data_ptr<int32_t>(), m,
                  out.data_ptr(), esz, cur_stream());
  return out;
}

// ------------------------------------------------------------------- group-by
std::tuple<torch::Tensor, torch::Tensor> hash_group_i64(torch::Tensor keys) {
  check_cuda(keys, "keys");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  int64_t n = keys.numel();
  auto opts32 = keys.options().dtype(torch::kInt32);
  if (n == 0)
    return {torch::empty({0}, opts32), torch::empty({0}, keys.options())};
  uint32_t tsize = next_pow2((uint32_t)std::max<int64_t>(n * 2, 64));
  auto table_keys = torch::empty({(int64_t)tsize}, keys.options());
  auto table_gids = torch::empty({(int64_t)tsize}, opts32);
  auto counter = torch::zeros({1}, opts32);
  auto gids = torch::empty({n}, opts32);
  auto st = cur_stream();
  launch_hash_build(keys.data_ptr<int64_t>(), n,
                    table_keys.data_ptr<int64_t>(),
                    table_gids.data_ptr<int32_t>(), tsize,
                    counter.data_ptr<int32_t>(), gids.data_ptr<int32_t>(), st);
  int64_t g = counter.item<int32_t>();  // sync to size the uniq output
  auto uniq = torch::empty({g}, keys.options());
  launch_hash_export(table_keys.data_ptr<int64_t>(),
                     table_gids.data_ptr<int32_t>(), tsize,
                     uniq.data_ptr<int64_t>(), st);
  return {gids, uniq};
}

torch::Tensor segment_reduce_f32(torch::Tensor vals, torch::Tensor gids,
                                 int64_t g, int64_t op) {
  check_cuda(vals, "vals");
  check_cuda(gids, "gids");
  TORCH_CHECK(vals.scalar_type() == torch::kFloat32, "vals must be f32");
  auto g32 = gids.scalar_type() == torch::kInt32 ? gids
                                                 : gids.to(torch::kInt32);
  auto out = torch::zeros({g}, vals.options());
  if (vals.numel() == 0) return out;
  torch::Tensor scratch;
  int32_t* scratch_ptr = nullptr;
  if (op != 0) {
    scratch = torch::empty({g}, vals.options().dtype(torch::kInt32));
    scratch_ptr = scratch.data_ptr<int32_t>();
  }
  launch_segment_reduce_f32(vals.data_ptr<float>(), g32.data_ptr<int32_t>(),
                            vals.numel(), (int)g, (int)op,
                            out.data_ptr<float>(), scratch_ptr, cur_stream());
  return out;
}

// ----------------------------------------------------------------------- join
std::tuple<torch::Tensor, torch::Tensor> join_inner_i64(torch::Tensor lk,
                                                        torch::Tensor rk) {
  check_cuda(lk, "left_keys");
  check_cuda(rk, "right_keys");
  TORCH_CHECK(lk.scalar_type() == torch::kInt64 &&
                  rk.scalar_type() == torch::kInt64,
              "join keys must be int64");
  int64_t nl = lk.numel(), nr = rk.numel();
  auto opts64 = lk.options();
  auto opts32 = lk.options().dtype(torch::kInt32);
  if (nl == 0 || nr == 0)
    return {torch::empty({0}, opts64), torch::empty({0}, opts64)};
  uint32_t tsize = next_pow2((uint32_t)std::max<int64_t>(nr * 2, 64));
  auto table_keys = torch::empty({(int64_t)tsize}, opts64);
  auto table_head = torch::empty({(int64_t)tsize}, opts32);
  auto next = torch::empty({nr}, opts32);
  auto st = cur_stream();
  launch_join_build(rk.data_ptr<int64_t>(), nr, table_keys.data_ptr<int64_t>(),
                    table_head.data_ptr<int32_t>(), next.data_ptr<int32_t>(),
                    tsize, st);
  auto counts = torch::empty({nl}, opts32);
  launch_join_probe_count(lk.data_ptr<int64_t>(), nl,
                          table_keys.data_ptr<int64_t>(),
                          table_head.data_ptr<int32_t>(),
                          next.data_ptr<int32_t>(), rk.data_ptr<int64_t>(),
                          tsize, counts.data_ptr<int32_t>(), st);
  auto [offs, total_t] = exscan(counts);
  int64_t total = total_t.item<int32_t>();
  auto l_out = torch::empty({total}, opts64);
  auto r_out = torch::empty({total}, opts64);
  if (total)
    launch_join_probe_emit(lk.data_ptr<int64_t>(), nl,
                           table_keys.data_ptr<int64_t>(),
                           table_head.data_ptr<int32_t>(),
                           next.data_ptr<int32_t>(), rk.data_ptr<int64_t>(),
                           tsize, offs.data_ptr<int32_t>(),
                           l_out.data_ptr<int64_t>(),
                           r_out.data_ptr<int64_t>(), st);
  return {l_out, r_out};
}

// ------------------------------------------------------------------ inference
torch::Tensor gemm_bf16(torch::Tensor A, torch::Tensor Bt,
                        c10::optional<torch::Tensor> bias, int64_t act) {
  check_cuda(A, "A");
  check_cuda(Bt, "Bt");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                  Bt.scalar_type() == torch::kBFloat16,
              "gemm_bf16 requires bf16 inputs");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2, "A [M,K], Bt [N,K]");
  int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 32 == 0 && K >= 32, "K must be a multiple of 32");
  auto C = torch::empty({M, N}, A.options());
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->scalar_type() == torch::kFloat32
                 ? bias->contiguous()
                 : bias->to(torch::kFloat32).contiguous();
    TORCH_CHECK(bias_f.numel() == N, "bias must be [N]");
    bias_ptr = bias_f.data_ptr<float>();
  }
  // measured dispatch (profiles/ r02/r06 GEMM tables): big square shapes →
  // 8-phase BM=256+swizzle; chip-filling mid shapes → 8-phase BM=128+swizzle;
  // everything else → the 128² single-buffer tile.
  int64_t t256 = ((M + 255) / 256) * ((N + 255) / 256);
  int64_t t128 = ((M + 127) / 128) * ((N + 255) / 256);
  // NOTE: BM128+swz wins STANDALONE at BERT shapes (+8-16%) but measured
  // ~2% slower in-context (L2-warm inputs; see profiles/ r06) — so only the
  // big-shape case dispatches to the pipelined kernel.
  (void)t128;
  int swz_code = -1;
  if (t256 >= 128 && K >= 1536 && N >= 2048)
    swz_code = 1;  // BM256 + swizzle
  int rc = swz_code >= 0
               ? launch_gemm_bf16_8p(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                     C.data_ptr(), (int)M, (int)N, (int)K,
                                     (int)act, swz_code, cur_stream())
               : -1;
  if (rc != 0) {
    // narrow outputs: the 128² grid can't fill 256 CUs (e.g. [8192,256] →
    // 128 blocks); the 64² tile quadruples the grid (profiles r2)
    int64_t t128grid = ((M + 127) / 128) * ((N + 127) / 128);
    if (t128grid < 208) {
      launch_gemm_bf16_skinny(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                              C.data_ptr(), (int)M, (int)N, (int)K,
                              (int)act, cur_stream());
    } else {
      // K%64==0 (BERT fc1/QKV/fc2 class): the swizzled BK=64 tile wins
      // 6-21% over BK=32 at every such shape measured (profiles r2
      // bench_gemm_k64; register staging + XOR chunk swizzle — the linear
      // layout's 128B-period LDS alias cost ~4e5 bank conflicts/dispatch).
      // Big squares already went to the 8-phase kernel above.
      int rc2 = (K % 64 == 0)
                    ? launch_gemm_bf16_k64s(A.data_ptr(), Bt.data_ptr(),
                                            bias_ptr, C.data_ptr(), (int)M,
                                            (int)N, (int)K, (int)act,
                                            cur_stream())
                    : -1;
      if (rc2 != 0)
        launch_gemm_bf16(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                         C.data_ptr(), (int)M, (int)N, (int)K, (int)act,
                         cur_stream());
    }
  }
  return C;
}

// scores = x[M,K]·w[K] + bias → f32 (the MLP head: N=1 GEMM + downcast in
// one launch instead of a hipBLASLt GEMV + copy kernel)
torch::Tensor gemv_bf16_f32(torch::Tensor x, torch::Tensor w, double bias) {
  check_cuda(x, "x");
  check_cuda(w, "w");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16, "bf16 required");
  TORCH_CHECK(x.dim() == 2 && w.numel() == x.size(1), "x [M,K], w [K]");
  auto out = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat32));
  launch_gemv_bf16_f32(x.data_ptr(), w.data_ptr(), (float)bias, x.size(0),
                       (int)x.size(1), out.data_ptr<float>(), cur_stream());
  return out;
}

// one fused launch generating every float column + the int64 key column of
// a synthetic batch; ctr[0] advances on-device so graph replays
// re-randomize (ops/stepgraph.py)
void gen_fields(torch::Tensor block, c10::optional<torch::Tensor> key,
                std::vector<double> lo, std::vector<double> width,
                int64_t key_lo, int64_t key_range, torch::Tensor ctr) {
  check_cuda(block, "block");
  TORCH_CHECK(block.dim() == 2 && block.scalar_type() == torch::kFloat32,
              "block must be [nf, n] f32");
  TORCH_CHECK(ctr.scalar_type() == torch::kInt64 && ctr.numel() == 2,
              "ctr must be int64[2] on device");
  int nf = (int)block.size(0);
  int64_t n = block.size(1);
  TORCH_CHECK((int)lo.size() == nf && (int)width.size() == nf);
  std::vector<float> lof(nf), wf(nf);
  for (int f = 0; f < nf; ++f) { lof[f] = (float)lo[f]; wf[f] = (float)width[f]; }
  int64_t* key_ptr = nullptr;
  if (key.has_value() && key->defined()) {
    check_cuda(*key, "key");
    TORCH_CHECK(key->numel() == n && key->scalar_type() == torch::kInt64);
    key_ptr = key->data_ptr<int64_t>();
  } else {
    key_range = 0;
  }
  launch_gen_fields(block.data_ptr<float>(), key_ptr, n, lof.data(),
                    wf.data(), nf, key_lo, key_range,
                    (unsigned long long*)ctr.data_ptr<int64_t>(),
                    cur_stream());
}

// One-launch generate+filter+compact(+featpack): see stepfused.hip
// genfiltpack_kernel. outs = compacted float columns; key_out optional;
// feats optional [cap, dpad] bf16. counts_ws int32[>=256] and bar int64[1]
// are persistent workspaces (the barrier counter is monotonic across graph
// replays — never reset them between steps).
void genfiltpack(std::vector<double> lo, std::vector<double> width,
                 int64_t key_lo, int64_t key_range, int64_t n, int64_t fidx,
                 int64_t op, double scalar,
                 std::vector<torch::Tensor> outs,
                 c10::optional<torch::Tensor> key_out,
                 c10::optional<torch::Tensor> feats,
                 torch::Tensor count_out, torch::Tensor counts_ws,
                 torch::Tensor bar, torch::Tensor ctr) {
  int nf = (int)outs.size();
  TORCH_CHECK(nf >= 1 && nf <= 32, "1..32 float columns");
  TORCH_CHECK((int)lo.size() == nf && (int)width.size() == nf);
  TORCH_CHECK(fidx >= 0 && fidx < nf, "fidx out of range");
  TORCH_CHECK(count_out.scalar_type() == torch::kInt32 &&
              count_out.numel() == 1);
  TORCH_CHECK(counts_ws.scalar_type() == torch::kInt32 &&
              counts_ws.numel() >= (n + 1023) / 1024);
  TORCH_CHECK(bar.scalar_type() == torch::kInt64 && bar.numel() >= 1);
  TORCH_CHECK(ctr.scalar_type() == torch::kInt64 && ctr.numel() == 2);
  std::vector<float> lof(nf), wf(nf);
  std::vector<float*> optr(nf);
  for (int f = 0; f < nf; ++f) {
    lof[f] = (float)lo[f];
    wf[f] = (float)width[f];
    check_cuda(outs[f], "out col");
    TORCH_CHECK(outs[f].numel() >= n &&
                outs[f].scalar_type() == torch::kFloat32);
    optr[f] = outs[f].data_ptr<float>();
  }
  int64_t* kptr = nullptr;
  if (key_out.has_value() && key_out->defined()) {
    TORCH_CHECK(key_out->numel() >= n &&
                key_out->scalar_type() == torch::kInt64);
    kptr = key_out->data_ptr<int64_t>();
  } else {
    key_range = 0;
  }
  void* fptr = nullptr;
  int dpad = 0;
  if (feats.has_value() && feats->defined()) {
    TORCH_CHECK(feats->dim() == 2 &&
                feats->scalar_type() == torch::kBFloat16 &&
                feats->size(0) >= n && feats->size(1) >= nf);
    fptr = feats->data_ptr();
    dpad = (int)feats->size(1);
  }
  int rc = launch_genfiltpack(
      lof.data(), wf.data(), nf, key_lo, key_range, n, (int)fidx, (int)op,
      (float)scalar, optr.data(), kptr, fptr, dpad,
      count_out.data_ptr<int32_t>(), counts_ws.data_ptr<int32_t>(),
      (unsigned long long*)bar.data_ptr<int64_t>(),
      (unsigned long long*)ctr.data_ptr<int64_t>(), cur_stream());
  TORCH_CHECK(rc == 0, "genfiltpack: batch too large for the in-kernel "
              "barrier (max 256K rows) — use the multi-kernel chain");
}

// gathered f32 feature columns → [n, kpad] bf16 MFMA operand in one launch
void featpack(std::vector<torch::Tensor> srcs, torch::Tensor out) {
  TORCH_CHECK(!srcs.empty() && srcs.size() <= 32, "1..32 feature columns");
  check_cuda(out, "out");
  TORCH_CHECK(out.dim() == 2 && out.scalar_type() == torch::kBFloat16,
              "out must be [n, kpad] bf16");
  int64_t n = out.size(0);
  int kpad = (int)out.size(1);
  bool f64 = srcs[0].scalar_type() == torch::kFloat64;
  if (f64) {
    std::vector<const double*> ptrs;
    for (auto& s : srcs) {
      check_cuda(s, "src");
      TORCH_CHECK(s.numel() >= n && s.scalar_type() == torch::kFloat64);
      ptrs.push_back(s.data_ptr<double>());
    }
    launch_featpack64(ptrs.data(), (int)ptrs.size(), kpad, n,
                      out.data_ptr(), cur_stream());
    return;
  }
  std::vector<const float*> ptrs;
  for (auto& s : srcs) {
    check_cuda(s, "src");
    TORCH_CHECK(s.numel() >= n && s.scalar_type() == torch::kFloat32);
    ptrs.push_back(s.data_ptr<float>());
  }
  launch_featpack(ptrs.data(), (int)ptrs.size(), kpad, n, out.data_ptr(),
                  cur_stream());
}

torch::Tensor gemm_bf16_variant(torch::Tensor A, torch::Tensor Bt,
                                c10::optional<torch::Tensor> bias,
                                int64_t act, int64_t variant) {
  // variant: 0 = 128² tile, 1 = 8-phase linear LDS, 2 = 8-phase + swizzle
  check_cuda(A, "A");
  check_cuda(Bt, "Bt");
  int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  auto C = torch::empty({M, N}, A.options());
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  if (variant == 0) {
    launch_gemm_bf16(A.data_ptr(), Bt.data_ptr(), bias_ptr, C.data_ptr(),
                     (int)M, (int)N, (int)K, (int)act, cur_stream());
  } else if (variant == 3) {
    int rc = launch_gemm_bf16_2p(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                 C.data_ptr(), (int)M, (int)N, (int)K,
                                 (int)act, cur_stream());
    TORCH_CHECK(rc == 0, "shape not supported by 2-phase kernel");
  } else if (variant == 6) {
    launch_gemm_bf16_skinny(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                            C.data_ptr(), (int)M, (int)N, (int)K, (int)act,
                            cur_stream());
  } else if (variant == 7) {
    int rc = launch_gemm_bf16_k64(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                  C.data_ptr(), (int)M, (int)N, (int)K,
                                  (int)act, cur_stream());
    TORCH_CHECK(rc == 0, "K must be a multiple of 64 for the BK=64 kernel");
  } else if (variant == 8) {
    int rc = launch_gemm_bf16_k64s(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                   C.data_ptr(), (int)M, (int)N, (int)K,
                                   (int)act, cur_stream());
    TORCH_CHECK(rc == 0, "K must be a multiple of 64 for the BK=64 kernel");
  } else if (variant == 10) {
    TORCH_CHECK(launch_gemm_bf16_k64p(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                      C.data_ptr(), (int)M, (int)N, (int)K,
                                      (int)act, cur_stream()) == 0,
                "k64p requires K % 64 == 0");
  } else if (variant == 11) {
    TORCH_CHECK(launch_gemm_bf16_k64w(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                      C.data_ptr(), (int)M, (int)N, (int)K,
                                      (int)act, cur_stream()) == 0,
                "k64w requires K % 64 == 0");
  } else if (variant == 12) {
    TORCH_CHECK(launch_gemm_bf16_k64s3(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                       C.data_ptr(), (int)M, (int)N, (int)K,
                                       (int)act, cur_stream()) == 0,
                "k64s3 requires K % 64 == 0");
  } else if (variant == 9) {
    int rc = launch_gemm_bf16_k64d(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                   C.data_ptr(), (int)M, (int)N, (int)K,
                                   (int)act, cur_stream());
    TORCH_CHECK(rc == 0, "K must be a multiple of 64 for the BK=64 kernel");
  } else {
    // 8-phase variants: 1=BM256, 2=BM256+swz, 4=BM128, 5=BM128+swz
    int swz_code = variant == 2 ? 1 : variant == 4 ? 2 : variant == 5 ? 3 : 0;
    int rc = launch_gemm_bf16_8p(A.data_ptr(), Bt.data_ptr(), bias_ptr,
                                 C.data_ptr(), (int)M, (int)N, (int)K,
                                 (int)act, swz_code, cur_stream());
    TORCH_CHECK(rc == 0, "shape not supported by 8-phase kernel");
  }
  return C;
}

torch::Tensor layernorm_bf16(torch::Tensor x, torch::Tensor gamma,
                             torch::Tensor beta, double eps,
                             c10::optional<torch::Tensor> residual) {
  check_cuda(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  int64_t n = x.size(-1);
  TORCH_CHECK(n % 8 == 0, "inner dim must be a multiple of 8");
  int64_t rows = x.numel() / n;
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto b = beta.to(torch::kFloat32).contiguous();
  auto out = torch::empty_like(x);
  const void* res_ptr = nullptr;
  if (residual.has_value() && residual->defined()) {
    TORCH_CHECK(residual->is_contiguous() &&
                residual->scalar_type() == torch::kBFloat16);
    res_ptr = residual->data_ptr();
  }
  launch_layernorm_bf16(x.data_ptr(), res_ptr, g.data_ptr<float>(),
                        b.data_ptr<float>(), out.data_ptr(), nullptr, rows,
                        (int)n, (float)eps, cur_stream());
  return out;
}

torch::Tensor softmax_bf16(torch::Tensor x, double scale) {
  check_cuda(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  int64_t n = x.size(-1);
  TORCH_CHECK(n % 8 == 0, "inner dim must be a multiple of 8");
  int64_t rows = x.numel() / n;
  auto out = torch::empty_like(x);
  launch_softmax_bf16(x.data_ptr(), out.data_ptr(), rows, (int)n,
                      (float)scale, cur_stream());
  return out;
}

// Host-mapped (pinned, device-visible) int32 buffer: the capture-safe filter
// writes its surviving-row count HERE over PCIe, so consuming a fused step is
// a stream sync + CPU load instead of a hipMemcpyDtoH launch + sync
// (flagship step is ~0.1 ms; the D2H readback was ~10 us of it).
// Returns (host_view, device_view) over the same allocation — keep both
// alive; freeing follows the host view.
std::vector<torch::Tensor> mapped_int32(int64_t n) {
  TORCH_CHECK(n > 0, "n must be positive");
  void* hp = nullptr;
  auto err = hipHostMalloc(&hp, n * sizeof(int32_t), hipHostMallocMapped);
  TORCH_CHECK(err == hipSuccess, "hipHostMalloc failed: ",
              hipGetErrorString(err));
  memset(hp, 0, n * sizeof(int32_t));
  void* dp = nullptr;
  err = hipHostGetDevicePointer(&dp, hp, 0);
  if (err != hipSuccess) {
    hipHostFree(hp);
    TORCH_CHECK(false, "hipHostGetDevicePointer failed: ",
                hipGetErrorString(err));
  }
  auto host = torch::from_blob(
      hp, {n}, [](void* p) { hipHostFree(p); },
      torch::TensorOptions().dtype(torch::kInt32));
  auto dev = torch::from_blob(
      dp, {n},
      torch::TensorOptions().dtype(torch::kInt32).device(torch::kCUDA));
  return {host, dev};
}

torch::Tensor bias_act_bf16(torch::Tensor x,
                            c10::optional<torch::Tensor> bias, int64_t act) {
  check_cuda(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  int64_t n = x.size(-1);
  int64_t rows = x.numel() / n;
  auto out = torch::empty_like(x);
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  launch_bias_act_bf16(x.data_ptr(), bias_ptr, out.data_ptr(), rows, (int)n,
                       (int)act, cur_stream());
  return out;
}

torch::Tensor attention_bf16(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                             double scale) {
  check_cuda(q, "q");
  check_cuda(k, "k");
  check_cuda(v, "v");
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16, "attention requires bf16");
  TORCH_CHECK(q.dim() == 4, "q must be [B,H,S,D]");
  int64_t B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto out = torch::empty_like(q);
  int rc = launch_attention_bf16(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                                 out.data_ptr(), (int)(B * H), (int)S, (int)D,
                                 (float)scale, cur_stream());
  TORCH_CHECK(rc == 0, "attention_bf16: unsupported shape S=", S, " D=", D);
  return out;
}

// qkv [B,S,3,H,D] (the QKV linear's output reshaped) → O [B,S,H*D];
// the kernel reads strided rows directly — no permute/contiguous copies.
torch::Tensor attention_qkv_bf16(torch::Tensor qkv, double scale,
                                 int64_t pad) {
  check_cuda(qkv, "qkv");
  TORCH_CHECK(qkv.scalar_type() == torch::kBFloat16, "qkv must be bf16");
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3, "qkv must be [B,S,3,H,D]");
  int64_t B = qkv.size(0), S = qkv.size(1), H = qkv.size(3), D = qkv.size(4);
  auto out = torch::empty({B, S, H * D}, qkv.options());
  int rc = launch_attention_qkv_bf16_pad(qkv.data_ptr(), out.data_ptr(),
                                         (int)B, (int)H, (int)S, (int)D,
                                         (float)scale, (int)pad,
                                         cur_stream());
  TORCH_CHECK(rc == 0, "attention_qkv_bf16: unsupported shape S=", S,
              " D=", D);
  return out;
}

std::vector<torch::Tensor> gather_columns(std::vector<torch::Tensor> cols,
                                          torch::Tensor idx) {
  check_cuda(idx, "idx");
  auto idx32 = idx.scalar_type() == torch::kInt32 ? idx : idx.to(torch::kInt32);
  int64_t m = idx32.numel();
  std::vector<torch::Tensor> outs;
  outs.reserve(cols.size());
  std::vector<const void*> src;
  std::vector<void*> dst;
  std::vector<int> esz;
  auto flush = [&]() {
    if (!src.empty() && m > 0)
      launch_gather_multi((int)src.size(), src.data(), dst.data(), esz.data(),
                          idx32.data_ptr<int32_t>(), m, cur_stream());
    src.clear(); dst.clear(); esz.clear();
  };
  for (auto& c : cols) {
    check_cuda(c, "col");
    auto out = torch::empty({m}, c.options());
    outs.push_back(out);
    src.push_back(c.data_ptr());
    dst.push_back(out.data_ptr());
    esz.push_back((int)c.element_size());
    if ((int)src.size() == 32) flush();
  }
  flush();
  return outs;
}

torch::Tensor radix_argsort(torch::Tensor keys, bool descending) {
  // Onesweep: 8-bit digits + decoupled lookback, keys ping-ponged alongside
  // the permutation (radix_sort.hip onesweep_* kernels). f32/i32 → 4 chained
  // passes, i64 → 8. The 4-bit two-kernel LSD this replaces measured ~6× off
  // rocPRIM at 10M keys (profiles r04 / VERDICT weak #7).
  check_cuda(keys, "keys");
  int64_t n = keys.numel();
  auto opts32 = keys.options().dtype(torch::kInt32);
  if (n == 0) return torch::empty({0}, opts32);
  TORCH_CHECK(n < (int64_t)1 << 30, "radix_argsort caps at 2^30 rows");
  auto st = cur_stream();
  bool wide;  // 64-bit key path
  torch::Tensor ka;
  if (keys.scalar_type() == torch::kFloat32) {
    wide = false;
    ka = torch::empty({n}, opts32);  // uint32 bits in an int32 tensor
    launch_f32_to_ordered(keys.data_ptr<float>(),
                          (uint32_t*)ka.data_ptr<int32_t>(), n,
                          descending ? 1 : 0, st);
  } else if (keys.scalar_type() == torch::kInt32) {
    wide = false;
    ka = torch::empty({n}, opts32);
    launch_i32_to_ordered(keys.data_ptr<int32_t>(),
                          (uint32_t*)ka.data_ptr<int32_t>(), n,
                          descending ? 1 : 0, st);
  } else if (keys.scalar_type() == torch::kInt64) {
    wide = true;
    ka = torch::empty({n}, keys.options().dtype(torch::kInt64));
    launch_i64_to_ordered(keys.data_ptr<int64_t>(),
                          (uint64_t*)ka.data_ptr<int64_t>(), n,
                          descending ? 1 : 0, st);
  } else {
    TORCH_CHECK(false, "radix_argsort supports f32/i64/i32 keys");
  }
  int npasses = wide ? 8 : 4;
  int nblocks = onesweep_nblocks(n);
  auto hist = torch::empty({(int64_t)256 * nblocks}, opts32);
  auto kb = torch::empty_like(ka);
  auto idx_a = torch::arange(n, opts32);
  auto idx_b = torch::empty({n}, opts32);
  for (int p = 0; p < npasses; ++p) {
    if (wide)
      launch_os_hist256_u64((const uint64_t*)ka.data_ptr<int64_t>(), n,
                            8 * p, hist.data_ptr<int32_t>(), nblocks, st);
    else
      launch_os_hist256_u32((const uint32_t*)ka.data_ptr<int32_t>(), n,
                            8 * p, hist.data_ptr<int32_t>(), nblocks, st);
    // one device scan turns bin-major block counts into global offsets
    auto inc = hist.cumsum(0, torch::kInt32);
    auto offs = (inc - hist).contiguous();
    if (wide)
      launch_onesweep_pass_u64((const uint64_t*)ka.data_ptr<int64_t>(),
                               idx_a.data_ptr<int32_t>(),
                               (uint64_t*)kb.data_ptr<int64_t>(),
                               idx_b.data_ptr<int32_t>(), n, 8 * p,
                               offs.data_ptr<int32_t>(), nblocks, st);
    else
      launch_onesweep_pass_u32((const uint32_t*)ka.data_ptr<int32_t>(),
                               idx_a.data_ptr<int32_t>(),
                               (uint32_t*)kb.data_ptr<int32_t>(),
                               idx_b.data_ptr<int32_t>(), n, 8 * p,
                               offs.data_ptr<int32_t>(), nblocks, st);
    std::swap(ka, kb);
    std::swap(idx_a, idx_b);
  }
  return idx_a;
}

// one-call C++ execution of `SELECT * FROM flow WHERE col OP literal`:
// fused compare+compact then one multi-column gather — a single host→device
// round trip instead of per-op Python dispatch (the native pipeline-executor
// path; reference's Pipeline is compiled Rust end to end).
std::tuple<std::vector<torch::Tensor>, int64_t> fused_filter_gather(
    std::vector<torch::Tensor> cols, int64_t filter_idx, int64_t op,
    double scalar) {
  TORCH_CHECK(filter_idx >= 0 && filter_idx < (int64_t)cols.size());
  auto col = cols[filter_idx];
  check_cuda(col, "filter col");
  int64_t n = col.numel();
  auto st = cur_stream();
  auto opts32 = col.options().dtype(torch::kInt32);
  if (n == 0) {
    std::vector<torch::Tensor> outs;
    for (auto& c : cols) outs.push_back(torch::empty({0}, c.options()));
    return {outs, 0};
  }
  int nblocks = filter_grid(n);
  auto counts = torch::empty({nblocks}, opts32);
  bool is_f32 = col.scalar_type() == torch::kFloat32;
  torch::Tensor col64;
  if (!is_f32) {
    col64 = col.scalar_type() == torch::kInt64 ? col : col.to(torch::kInt64);
  }
  if (is_f32)
    launch_filter_count_f32(col.data_ptr<float>(), n, (int)op, (float)scalar,
                            counts.data_ptr<int32_t>(), st);
  else
    launch_filter_count_i64(col64.data_ptr<int64_t>(), n, (int)op,
                            (int64_t)scalar, counts.data_ptr<int32_t>(), st);
  auto [offs, total_t] = exscan(counts);
  int64_t total = total_t.item<int32_t>();  // the ONE host sync
  auto idx = torch::empty({total}, opts32);
  std::vector<torch::Tensor> outs;
  outs.reserve(cols.size());
  if (total) {
    if (is_f32)
      launch_filter_scatter_f32(col.data_ptr<float>(), n, (int)op,
                                (float)scalar, offs.data_ptr<int32_t>(),
                                idx.data_ptr<int32_t>(), st);
    else
      launch_filter_scatter_i64(col64.data_ptr<int64_t>(), n, (int)op,
                                (int64_t)scalar, offs.data_ptr<int32_t>(),
                                idx.data_ptr<int32_t>(), st);
    std::vector<const void*> src;
    std::vector<void*> dst;
    std::vector<int> esz;
    auto flush = [&]() {
      if (!src.empty())
        launch_gather_multi((int)src.size(), src.data(), dst.data(),
                            esz.data(), idx.data_ptr<int32_t>(), total, st);
      src.clear(); dst.clear(); esz.clear();
    };
    for (auto& c : cols) {
      auto out = torch::empty({total}, c.options());
      outs.push_back(out);
      src.push_back(c.data_ptr());
      dst.push_back(out.data_ptr());
      esz.push_back((int)c.element_size());
      if ((int)src.size() == 32) flush();
    }
    flush();
  } else {
    for (auto& c : cols) outs.push_back(torch::empty({0}, c.options()));
  }
  return {outs, total};
}

// hipGraph-capturable variant of fused_filter_gather: writes compacted rows
// into PREALLOCATED padded outputs and the row count into a device int32 —
// zero host syncs, so generate→filter→infer captures as ONE graph
// (the flagship step was host-dispatch-bound at batch 8192).
void filter_gather_capture(std::vector<torch::Tensor> cols,
                           int64_t filter_idx, int64_t op, double scalar,
                           std::vector<torch::Tensor> outs,
                           torch::Tensor count_out) {
  TORCH_CHECK(filter_idx >= 0 && filter_idx < (int64_t)cols.size());
  TORCH_CHECK(outs.size() == cols.size(), "outs must match cols");
  TORCH_CHECK(count_out.scalar_type() == torch::kInt32 &&
              count_out.numel() == 1, "count_out must be int32[1] on device");
  auto col = cols[filter_idx];
  check_cuda(col, "filter col");
  int64_t n = col.numel();
  auto st = cur_stream();
  TORCH_CHECK(n > 0, "capture path requires a non-empty static batch");
  int nblocks = filter_grid(n);
  auto counts = torch::empty({nblocks}, col.options().dtype(torch::kInt32));
  bool is_f32 = col.scalar_type() == torch::kFloat32;
  TORCH_CHECK(is_f32 || col.scalar_type() == torch::kInt64,
              "capture filter col must be f32 or i64");
  if (is_f32)
    launch_filter_count_f32(col.data_ptr<float>(), n, (int)op, (float)scalar,
                            counts.data_ptr<int32_t>(), st);
  else
    launch_filter_count_i64(col.data_ptr<int64_t>(), n, (int)op,
                            (int64_t)scalar, counts.data_ptr<int32_t>(), st);
  torch::Tensor offs;
  if (nblocks <= 1024) {
    // single-block scan writes offsets AND the device total in one launch
    // (replaces torch cumsum + sub + copy_ — 3 kernels → 1 in the graph)
    offs = torch::empty({nblocks}, col.options().dtype(torch::kInt32));
    launch_scan_counts(counts.data_ptr<int32_t>(), nblocks,
                       offs.data_ptr<int32_t>(),
                       count_out.data_ptr<int32_t>(), st);
  } else {
    auto [offs2, total_t] = exscan(counts);
    offs = offs2;
    count_out.copy_(total_t, /*non_blocking=*/true);
  }
  auto idx = torch::empty({n}, col.options().dtype(torch::kInt32));
  if (is_f32)
    launch_filter_scatter_f32(col.data_ptr<float>(), n, (int)op,
                              (float)scalar, offs.data_ptr<int32_t>(),
                              idx.data_ptr<int32_t>(), st);
  else
    launch_filter_scatter_i64(col.data_ptr<int64_t>(), n, (int)op,
                              (int64_t)scalar, offs.data_ptr<int32_t>(),
                              idx.data_ptr<int32_t>(), st);
  std::vector<const void*> src;
  std::vector<void*> dst;
  std::vector<int> esz;
  auto flush = [&]() {
    if (!src.empty())
      launch_gather_multi_dyn((int)src.size(), src.data(), dst.data(),
                              esz.data(), idx.data_ptr<int32_t>(),
                              count_out.data_ptr<int32_t>(), n, st);
    src.clear(); dst.clear(); esz.clear();
  };
  for (size_t c = 0; c < cols.size(); ++c) {
    TORCH_CHECK(outs[c].numel() >= n &&
                outs[c].element_size() == cols[c].element_size(),
                "out buffer too small or wrong dtype");
    src.push_back(cols[c].data_ptr());
    dst.push_back(outs[c].data_ptr());
    esz.push_back((int)cols[c].element_size());
    if ((int)src.size() == 32) flush();
  }
  flush();
}

// one-call `SELECT key, count(*), AGG(col)… FROM flow WHERE fcol OP lit
// GROUP BY key`: fused filter → key/value gather → LDS hash group →
// segment reductions, chained in C++ with exactly TWO host syncs (surviving
// row count, group count). The Python-op version of this chain is
// host-dispatch-bound at stream batch sizes (BASELINE config 2 @8192).
// val_ops: 0=sum 1=min 2=max per value column. Returns
// (unique_keys, group_counts_f32, [reduced…]).
std::tuple<torch::Tensor, torch::Tensor, std::vector<torch::Tensor>>
fused_filter_agg(torch::Tensor key, torch::Tensor filter_col, int64_t op,
                 double scalar, std::vector<torch::Tensor> val_cols,
                 std::vector<int64_t> val_ops) {
  check_cuda(key, "key");
  check_cuda(filter_col, "filter_col");
  TORCH_CHECK(key.scalar_type() == torch::kInt64, "key must be int64");
  TORCH_CHECK(filter_col.scalar_type() == torch::kFloat32,
              "filter col must be f32");
  TORCH_CHECK(val_cols.size() == val_ops.size());
  int64_t n = key.numel();
  auto st = cur_stream();
  auto opts32 = key.options().dtype(torch::kInt32);
  std::vector<torch::Tensor> cols = {key, filter_col};
  for (auto& v : val_cols) {
    check_cuda(v, "val col");
    TORCH_CHECK(v.scalar_type() == torch::kFloat32, "val cols must be f32");
    cols.push_back(v);
  }
  auto [outs, total] = fused_filter_gather(cols, 1, op, scalar);
  auto gkey = outs[0];
  if (total == 0) {
    std::vector<torch::Tensor> empt;
    for (size_t i = 0; i < val_cols.size(); ++i)
      empt.push_back(torch::empty({0}, filter_col.options()));
    return {torch::empty({0}, key.options()),
            torch::empty({0}, filter_col.options()), empt};
  }
  auto [gids, uniq] = hash_group_i64(gkey);
  int64_t g = uniq.numel();
  // counts via the f32 LDS-buffered segment sum (exact < 2^24 per group)
  auto ones = torch::ones({total}, filter_col.options());
  auto counts = segment_reduce_f32(ones, gids, g, 0);
  std::vector<torch::Tensor> reduced;
  for (size_t i = 0; i < val_cols.size(); ++i)
    reduced.push_back(segment_reduce_f32(outs[2 + i], gids, g,
                                         val_ops[i]));
  return {uniq, counts, reduced};
}

// Capture-safe GROUP BY: padded key/value columns + the filter's device row
// count in, padded (g_cap) group table out, group count written to
// gcount_out (host-mapped int32 — ONE CPU read per step, zero D2H copies).
// Records into a hipGraph: static grids, no host syncs; all workspaces are
// allocated here (graph-pool when called during capture). Reference analog:
// DataFusion AggregateExec (processor/sql.rs execute_query) as a replayable
// device-side program.
std::tuple<torch::Tensor, torch::Tensor, std::vector<torch::Tensor>>
hash_agg_capture(torch::Tensor keys, torch::Tensor nrow,
                 std::vector<torch::Tensor> vals, std::vector<int64_t> ops,
                 int64_t table_size, int64_t g_cap,
                 c10::optional<torch::Tensor> gcount_out) {
  check_cuda(keys, "keys");
  check_cuda(nrow, "nrow");
  TORCH_CHECK(keys.scalar_type() == torch::kInt64, "keys must be int64");
  TORCH_CHECK(nrow.scalar_type() == torch::kInt32 && nrow.numel() == 1,
              "nrow must be device int32[1]");
  TORCH_CHECK((table_size & (table_size - 1)) == 0, "table_size pow2");
  TORCH_CHECK(g_cap >= 1 && g_cap <= 2048, "g_cap must be in [1, 2048]");
  TORCH_CHECK(vals.size() == ops.size());
  TORCH_CHECK(vals.size() <= 4, "hash_agg_capture supports up to 4 "
              "reduction columns");
  int64_t n_cap = keys.numel();
  auto st = cur_stream();
  auto opts32 = keys.options().dtype(torch::kInt32);
  auto optsf = keys.options().dtype(torch::kFloat32);
  auto table_keys = torch::empty({table_size}, keys.options());
  auto table_gids = torch::empty({table_size}, opts32);
  auto counter = torch::empty({1}, opts32);
  auto gids = torch::empty({std::max<int64_t>(n_cap, 1)}, opts32);
  auto counts = torch::empty({g_cap}, optsf);
  auto counts_i64 = torch::empty({g_cap}, keys.options());
  auto uniq = torch::zeros({g_cap}, keys.options());
  int nv = (int)vals.size();
  std::vector<torch::Tensor> red;
  std::vector<const float*> vptr;
  std::vector<float*> rptr;
  std::vector<int32_t*> mptr;
  std::vector<int> opsi;
  std::vector<torch::Tensor> scratch_keep;
  for (int i = 0; i < nv; ++i) {
    check_cuda(vals[i], "val");
    TORCH_CHECK(vals[i].scalar_type() == torch::kFloat32,
                "val cols must be f32");
    red.push_back(torch::empty({g_cap}, optsf));
    vptr.push_back(vals[i].data_ptr<float>());
    rptr.push_back(red.back().data_ptr<float>());
    if (ops[i] != 0) {
      scratch_keep.push_back(torch::empty({g_cap}, opts32));
      mptr.push_back(scratch_keep.back().data_ptr<int32_t>());
    } else {
      mptr.push_back(nullptr);
    }
    opsi.push_back((int)ops[i]);
  }
  int32_t* gout = nullptr;
  if (gcount_out.has_value() && gcount_out->defined()) {
    TORCH_CHECK(gcount_out->scalar_type() == torch::kInt32 &&
                gcount_out->numel() == 1);
    gout = gcount_out->data_ptr<int32_t>();
  }
  launch_hash_agg_capture(
      keys.data_ptr<int64_t>(), nrow.data_ptr<int32_t>(), n_cap,
      table_keys.data_ptr<int64_t>(), table_gids.data_ptr<int32_t>(),
      (uint32_t)table_size, counter.data_ptr<int32_t>(),
      gids.data_ptr<int32_t>(), counts.data_ptr<float>(), (int)g_cap,
      vptr.data(), opsi.data(), rptr.data(), mptr.data(), nv,
      uniq.data_ptr<int64_t>(), counts_i64.data_ptr<int64_t>(), gout, st);
  return {uniq, counts_i64, red};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
           std::vector<std::tuple<torch::Tensor, torch::Tensor>>,
           torch::Tensor>
json_decode(torch::Tensor data, torch::Tensor offsets,
            std::vector<std::string> names, std::vector<int64_t> kind,
            std::vector<int64_t> slot, int64_t n_int, int64_t n_float,
            int64_t n_str) {
  check_cuda(data, "data");
  check_cuda(offsets, "offsets");
  int64_t n = offsets.numel() - 1;
  int nf = (int)names.size();
  std::vector<char> packed(nf * 24, 0);
  std::vector<int> nl(nf), kd(nf), sl(nf), par(nf);
  // dotted names ("user.id") address one level of nesting: collect the
  // distinct parent segments, scope each field to its parent index
  std::vector<std::string> parents;
  for (int f = 0; f < nf; ++f) {
    std::string leaf = names[f];
    par[f] = -1;
    auto dot = leaf.find('.');
    if (dot != std::string::npos) {
      std::string parent = leaf.substr(0, dot);
      leaf = leaf.substr(dot + 1);
      TORCH_CHECK(leaf.find('.') == std::string::npos,
                  "json paths support one nesting level: ", names[f]);
      int pi = -1;
      for (size_t q = 0; q < parents.size(); ++q)
        if (parents[q] == parent) pi = (int)q;
      if (pi < 0) {
        parents.push_back(parent);
        pi = (int)parents.size() - 1;
      }
      par[f] = pi;
    }
    TORCH_CHECK(leaf.size() < 24, "json field name too long");
    memcpy(&packed[f * 24], leaf.data(), leaf.size());
    nl[f] = (int)leaf.size();
    kd[f] = (int)kind[f];
    sl[f] = (int)slot[f];
  }
  TORCH_CHECK(parents.size() <= 8, "too many nested parents (max 8)");
  int np = (int)parents.size();
  std::vector<char> ppacked(std::max(np, 1) * 24, 0);
  std::vector<int> pl(std::max(np, 1), 0);
  for (int q = 0; q < np; ++q) {
    TORCH_CHECK(parents[q].size() < 24, "json parent name too long");
    memcpy(&ppacked[q * 24], parents[q].data(), parents[q].size());
    pl[q] = (int)parents[q].size();
  }
  auto out_f = torch::zeros({std::max<int64_t>(n_float, 1),
                             std::max<int64_t>(n, 1)},
                            data.options().dtype(torch::kFloat64));
  auto out_i = torch::zeros({std::max<int64_t>(n_int, 1),
                             std::max<int64_t>(n, 1)},
                            data.options().dtype(torch::kInt64));
  auto str_start = torch::zeros({std::max<int64_t>(n_str, 1),
                                 std::max<int64_t>(n, 1)},
                                data.options().dtype(torch::kInt64));
  auto str_ulen = torch::zeros({std::max<int64_t>(n_str, 1),
                                std::max<int64_t>(n, 1)},
                               data.options().dtype(torch::kInt32));
  auto found = torch::zeros({std::max(nf, 1), std::max<int64_t>(n, 1)},
                            data.options().dtype(torch::kUInt8));
  auto err = torch::zeros({1}, data.options().dtype(torch::kInt32));
  auto found_count = torch::zeros({std::max(nf, 1)},
                                  data.options().dtype(torch::kInt32));
  if (n > 0)
    launch_json_decode(data.data_ptr<uint8_t>(), offsets.data_ptr<int64_t>(),
                       n, nf, packed.data(), nl.data(), kd.data(), sl.data(),
                       par.data(), np, ppacked.data(), pl.data(),
                       out_f.data_ptr<double>(), out_i.data_ptr<int64_t>(),
                       str_start.data_ptr<int64_t>(),
                       str_ulen.data_ptr<int32_t>(),
                       found.data_ptr<uint8_t>(), err.data_ptr<int32_t>(),
                       found_count.data_ptr<int32_t>(),
                       // wave-per-doc parse pays off only when string
                       // scans dwarf the per-doc state machine (which the
                       // thread-per-doc kernel amortizes 64 docs/wave):
                       // measured crossover is in the multi-KB range
                       (int)(n > 0 && data.numel() / n >= 2048),
                       cur_stream());
  // string copy-out: per field, offsets = exclusive cumsum of unescaped
  // lengths; one host sync for ALL totals at once, then one copy kernel per
  // string field (ulen of absent docs is 0 → empty strings, validity=found)
  std::vector<std::tuple<torch::Tensor, torch::Tensor>> strings;
  // ONE host readback for err + per-field all-valid + string totals — the
  // previous shape (err.item + validity .cpu + totals .to(CPU)) cost three
  // stream drains per batch and floored small-batch decode at ~0.2 ms
  std::vector<torch::Tensor> sumv;
  sumv.push_back(err.to(torch::kInt64));
  // all_valid[f] ⇔ found_count[f] == n (counts accumulated in-kernel —
  // a found.min(dim=1) reduce cost ~40 µs/batch)
  sumv.push_back((found_count.to(torch::kInt64) ==
                  torch::full({std::max(nf, 1)}, n,
                              torch::dtype(torch::kInt64)
                                  .device(data.device())))
                     .to(torch::kInt64));
  std::vector<torch::Tensor> offs(std::max<int64_t>(n_str, 0));
  if (n_str > 0 && n > 0) {
    for (int64_t s = 0; s < n_str; ++s) {
      offs[s] = exclusive_offsets(str_ulen[s].contiguous());
      sumv.push_back(offs[s].narrow(0, n, 1));
    }
  }
  auto summary = torch::cat(sumv).cpu();  // the one device sync
  if (n_str > 0 && n > 0) {
    auto* tot = summary.data_ptr<int64_t>() + 1 + std::max(nf, 1);
    for (int64_t s = 0; s < n_str; ++s) {
      auto out = torch::empty({std::max<int64_t>(tot[s], 1)},
                              data.options().dtype(torch::kUInt8));
      // found row for this slot: find the field index with this slot
      int frow = 0;
      for (int f = 0; f < nf; ++f)
        if (kd[f] == 2 && sl[f] == (int)s) frow = f;
      if (tot[s] > 0) {
        // wave-per-doc for long strings (coalesced fast path); the
        // thread-per-doc kernel stays ahead when strings are tiny
        if (tot[s] / n >= 32)
          launch_json_copy_strings_wave(
              data.data_ptr<uint8_t>(), str_start[s].data_ptr<int64_t>(),
              offs[s].data_ptr<int64_t>(), found[frow].data_ptr<uint8_t>(),
              n, data.numel(), out.data_ptr<uint8_t>(), cur_stream());
        else
          launch_json_copy_strings(
              data.data_ptr<uint8_t>(), str_start[s].data_ptr<int64_t>(),
              offs[s].data_ptr<int64_t>(), found[frow].data_ptr<uint8_t>(),
              n, data.numel(), out.data_ptr<uint8_t>(), cur_stream());
      }
      strings.emplace_back(out.slice(0, 0, tot[s]), offs[s]);
    }
  }
  return {out_f, out_i, found, err, strings, summary};
}

// gather rows of a binary column: (data, offsets, idx) -> (out_data,
// out_offsets). Device prefix-sum for the new offsets + span-copy kernel;
// ONE host sync (the total, needed for allocation).
std::tuple<torch::Tensor, torch::Tensor> take_binary(torch::Tensor data,
                                                     torch::Tensor offsets,
                                                     torch::Tensor idx) {
  check_cuda(data, "data");
  check_cuda(offsets, "offsets");
  check_cuda(idx, "idx");
  int64_t n = offsets.numel() - 1;
  auto idx64 = idx.scalar_type() == torch::kInt64 ? idx : idx.to(torch::kInt64);
  int64_t m = idx64.numel();
  auto lens_all = offsets.slice(0, 1, n + 1) - offsets.slice(0, 0, n);
  auto lens = lens_all.index_select(0, idx64).to(torch::kInt32);
  auto starts = offsets.slice(0, 0, n).index_select(0, idx64).contiguous();
  auto new_off = exclusive_offsets(lens.contiguous());
  int64_t total = m ? new_off.narrow(0, m, 1).item<int64_t>() : 0;
  auto out = torch::empty({std::max<int64_t>(total, 1)},
                          data.options().dtype(torch::kUInt8));
  if (total > 0)
    launch_proto_copy_bytes(data.data_ptr<uint8_t>(),
                            starts.data_ptr<int64_t>(),
                            new_off.data_ptr<int64_t>(), m,
                            out.data_ptr<uint8_t>(), cur_stream());
  return {out.slice(0, 0, total), new_off};
}

torch::Tensor bytes_match(torch::Tensor data, torch::Tensor offsets,
                          std::string needle, int64_t mode) {
  check_cuda(data, "data");
  check_cuda(offsets, "offsets");
  TORCH_CHECK(needle.size() <= 64, "LIKE needle too long for device match");
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, data.options().dtype(torch::kBool));
  if (n > 0)
    launch_bytes_match(data.data_ptr<uint8_t>(), offsets.data_ptr<int64_t>(),
                       n, (const uint8_t*)needle.data(), (int)needle.size(),
                       (int)mode, out.data_ptr<bool>(), cur_stream());
  return out;
}

torch::Tensor bytes_hash(torch::Tensor data, torch::Tensor offsets) {
  check_cuda(data, "data");
  check_cuda(offsets, "offsets");
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kInt64));
  if (n > 0)
    launch_bytes_hash(data.data_ptr<uint8_t>(), offsets.data_ptr<int64_t>(),
                      n, out.data_ptr<int64_t>(), cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           std::vector<std::tuple<torch::Tensor, torch::Tensor>>,
           torch::Tensor>
proto_decode(
    torch::Tensor data, torch::Tensor offsets,
    std::vector<int64_t> fno, std::vector<int64_t> kind,
    std::vector<int64_t> is_float, std::vector<int64_t> slot,
    int64_t n_int, int64_t n_float, int64_t n_str, bool capture) {
  check_cuda(data, "data");
  check_cuda(offsets, "offsets");
  TORCH_CHECK(data.scalar_type() == torch::kUInt8);
  TORCH_CHECK(offsets.scalar_type() == torch::kInt64);
  int64_t n = offsets.numel() - 1;
  int nf = (int)fno.size();
  std::vector<int> f(nf), k(nf), isf(nf), sl(nf);
  for (int i = 0; i < nf; ++i) {
    f[i] = (int)fno[i]; k[i] = (int)kind[i];
    isf[i] = (int)is_float[i]; sl[i] = (int)slot[i];
  }
  auto out_i = torch::zeros({std::max<int64_t>(n_int, 1), std::max<int64_t>(n, 1)},
                            data.options().dtype(torch::kInt64));
  auto out_f = torch::zeros({std::max<int64_t>(n_float, 1), std::max<int64_t>(n, 1)},
                            data.options().dtype(torch::kFloat64));
  auto str_start = torch::zeros({std::max<int64_t>(n_str, 1),
                                 std::max<int64_t>(n, 1)},
                                data.options().dtype(torch::kInt64));
  auto str_len = torch::zeros({std::max<int64_t>(n_str, 1),
                               std::max<int64_t>(n, 1)},
                              data.options().dtype(torch::kInt32));
  auto err = torch::zeros({1}, data.options().dtype(torch::kInt32));
  if (n > 0)
    launch_proto_decode(data.data_ptr<uint8_t>(), offsets.data_ptr<int64_t>(),
                        n, nf, f.data(), k.data(), isf.data(), sl.data(),
                        out_i.data_ptr<int64_t>(), out_f.data_ptr<double>(),
                        str_start.data_ptr<int64_t>(),
                        str_len.data_ptr<int32_t>(), err.data_ptr<int32_t>(),
                        cur_stream());
  // string/bytes copy-out (proto3 missing field → len 0 → empty value);
  // err + totals ride ONE host readback (JSON decode learned the same —
  // each extra sync floored small-batch decode)
  std::vector<std::tuple<torch::Tensor, torch::Tensor>> strings;
  if (capture) {
    // hipGraph capture: no host syncs allowed — string fields (which need
    // a totals readback for allocation) are unsupported; the caller
    // validates err outside the capture
    TORCH_CHECK(n_str == 0, "capture-mode proto decode cannot have strings");
    return {out_i, out_f, err, strings, torch::empty({0})};
  }
  std::vector<torch::Tensor> sumv = {err.to(torch::kInt64)};
  std::vector<torch::Tensor> offs(std::max<int64_t>(n_str, 0));
  if (n_str > 0 && n > 0) {
    for (int64_t s = 0; s < n_str; ++s) {
      offs[s] = exclusive_offsets(str_len[s].contiguous());
      sumv.push_back(offs[s].narrow(0, n, 1));
    }
  }
  auto summary = torch::cat(sumv).cpu();
  if (n_str > 0 && n > 0) {
    auto* tot = summary.data_ptr<int64_t>() + 1;
    for (int64_t s = 0; s < n_str; ++s) {
      auto out = torch::empty({std::max<int64_t>(tot[s], 1)},
                              data.options().dtype(torch::kUInt8));
      if (tot[s] > 0)
        launch_proto_copy_bytes(data.data_ptr<uint8_t>(),
                                str_start[s].data_ptr<int64_t>(),
                                offs[s].data_ptr<int64_t>(), n,
                                out.data_ptr<uint8_t>(), cur_stream());
      strings.emplace_back(out.slice(0, 0, tot[s]), offs[s]);
    }
  }
  return {out_i, out_f, err, strings, summary};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "arkflow_amd gfx950 native kernels";
  m.def("exclusive_offsets", &exclusive_offsets);
  m.def("take_binary", &take_binary);
  m.def("bytes_match", &bytes_match);
  m.def("attention_qkv_bf16", &attention_qkv_bf16, py::arg("qkv"),
        py::arg("scale"), py::arg("pad") = 8);
  m.def("mask_to_indices", &mask_to_indices);
  m.def("filter_cmp_scalar", &filter_cmp_scalar);
  m.def("gather", &gather);
  m.def("hash_group_i64", &hash_group_i64);
  m.def("segment_reduce_f32", &segment_reduce_f32);
  m.def("join_inner_i64", &join_inner_i64);
  m.def("gemm_bf16", &gemm_bf16, py::arg("A"), py::arg("Bt"),
        py::arg("bias") = py::none(), py::arg("act") = 0);
  m.def("layernorm_bf16", &layernorm_bf16, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("eps") = 1e-5,
        py::arg("residual") = py::none());
  m.def("softmax_bf16", &softmax_bf16, py::arg("x"), py::arg("scale") = 1.0);
  m.def("mapped_int32", &mapped_int32, py::arg("n"));
  m.def("bias_act_bf16", &bias_act_bf16, py::arg("x"),
        py::arg("bias") = py::none(), py::arg("act") = 0);
  m.def("attention_bf16", &attention_bf16, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("scale"));
  m.def("proto_decode", &proto_decode, py::arg("data"),
        py::arg("offsets"), py::arg("fno"), py::arg("kind"),
        py::arg("is_float"), py::arg("slot"), py::arg("n_int"),
        py::arg("n_float"), py::arg("n_str"), py::arg("capture") = false);
  m.def("gather_columns", &gather_columns);
  m.def("gemm_bf16_variant", &gemm_bf16_variant);
  m.def("bytes_hash", &bytes_hash);
  m.def("json_decode", &json_decode);
  m.def("fused_filter_gather", &fused_filter_gather);
  m.def("genfiltpack", &genfiltpack, py::arg("lo"), py::arg("width"),
        py::arg("key_lo"), py::arg("key_range"), py::arg("n"),
        py::arg("fidx"), py::arg("op"), py::arg("scalar"), py::arg("outs"),
        py::arg("key_out"), py::arg("feats"), py::arg("count_out"),
        py::arg("counts_ws"), py::arg("bar"), py::arg("ctr"));
  m.def("fused_filter_agg", &fused_filter_agg);
  m.def("hash_agg_capture", &hash_agg_capture, py::arg("keys"),
        py::arg("nrow"), py::arg("vals"), py::arg("ops"),
        py::arg("table_size"), py::arg("g_cap"),
        py::arg("gcount_out") = c10::nullopt);
  m.def("filter_gather_capture", &filter_gather_capture);
  m.def("gemv_bf16_f32", &gemv_bf16_f32);
  m.def("gen_fields", &gen_fields);
  m.def("featpack", &featpack);
  m.def("radix_argsort", &radix_argsort, py::arg("keys"), py::arg("descending") = false);
}
