// Python bindings for the parallax_amd gfx950 kernels (torch extension).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <cstdlib>

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) \
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")

extern "C" {
void launch_rmsnorm(void*, void*, const void*, float, int, int, hipStream_t);
void launch_fused_add_rmsnorm(void*, void*, const void*, float, int, int,
                              hipStream_t);
void launch_rope(void*, void*, const int*, const float*, int, int, int, int,
                 int, bool, int64_t, int64_t, hipStream_t);
void launch_rope_and_cache(void*, const void*, const void*, void*, void*,
                           const int*, const float*, const int64_t*, int, int,
                           int, int, int, int, bool, bool, float, float,
                           int64_t, int64_t, int64_t, hipStream_t);
void launch_reshape_and_cache(const void*, const void*, void*, void*,
                              const int64_t*, int, int, int, int, hipStream_t);
void launch_mla_reshape_and_cache(const void*, const void*, void*,
                                  const int64_t*, int, int, int, int, bool,
                                  float, hipStream_t);
void launch_act_and_mul(void*, const void*, int64_t, int, bool, hipStream_t);
void launch_paged_attention_decode(void*, const void*, const void*, const void*,
                                   const int*, const int*, int, int, int, int,
                                   int, int, int64_t, float, int, float,
                                   const float*, int, int, float*, float*,
                                   hipStream_t, bool*);
void launch_paged_decode_mfma(void*, const void*, const void*, const void*,
                              const int*, const int*, int, int, int, int, int,
                              int, int64_t, float, int, float, const float*,
                              int, int, float*, float*, bool, float, float,
                              hipStream_t, bool*);
void launch_paged_attention_reduce(void*, const float*, const float*, int, int,
                                   int, int, const float*, hipStream_t);
void launch_mla_paged_attention_decode(void*, const void*, const void*,
                                       const void*, const int*, const int*,
                                       int, int, int, int, int, int, float,
                                       int, int, float*, float*, const int*,
                                       int, bool, float, hipStream_t, bool*);
void launch_msa_paged_attention_decode(void*, const void*, const void*,
                                       const void*, const int*, const int*,
                                       const int*, int, int, int, int, int,
                                       int, int, int, int64_t, float,
                                       hipStream_t, bool*);
void launch_build_moe_tiles(int*, int*, const int*, int, int, hipStream_t);
void launch_moe_gate_up(void*, const void*, const void*, const int*,
                        const int*, const int64_t*, const int*, int, int, int,
                        int, int, bool, float, const void*, hipStream_t);
void launch_moe_down(void*, const void*, const void*, const float*, const int*,
                     const int*, const int64_t*, const int*, int, int, int,
                     int, int, hipStream_t);
void launch_dsa_indexer_scores(float*, const void*, const void*, const float*,
                               const int*, const int*, int, int, int, int,
                               int, int, hipStream_t, bool*);
void launch_store_indexer_cache(void*, const void*, const int64_t*, int,
                                int64_t, int, hipStream_t);
void launch_msa_block_scores(float*, const void*, const void*, const int*,
                             const int*, int, int, int, int, int, int, int,
                             int, hipStream_t, bool*);
void launch_msa_topk_tokens(int64_t*, const float*, const int*, int, int, int,
                            int, int, int, int, hipStream_t);
void launch_quantize_fp8_rows(void*, float*, const void*, int, int,
                              hipStream_t);
void launch_moe_gate_up_fp8(void*, const void*, const float*, const void*,
                            const float*, const int*, const int*,
                            const int64_t*, const int*, int, int, int, int,
                            int, bool, float, hipStream_t);
void launch_moe_down_fp8(void*, const void*, const float*, const void*,
                         const float*, const float*, const int*, const int*,
                         const int64_t*, const int*, int, int, int, int, int,
                         hipStream_t);
void launch_sample_gumbel(int64_t*, const float*, const float*,
                          const uint8_t*, uint64_t, int, int, hipStream_t);
void launch_skinny_gemm(void*, const void*, const void*, const void*, int,
                        int, int, int64_t, int64_t, hipStream_t, bool*);
void launch_prefill_attention(void*, const void*, const void*, const void*,
                              const int*, const int*, const int*, const int*,
                              const int*, int, int, int, int, int, int, int64_t,
                              float, int, float, const float*, bool, float,
                              float, hipStream_t, bool*);
}

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// [T, H, D] view whose rows may live inside a wider fused tensor (e.g. the QKV
// GEMM output): inner two dims must be dense, the token stride may be larger.
static bool is_fp8(const torch::Tensor& t) {
  return t.scalar_type() == at::kFloat8_e4m3fn;
}

static int64_t row_stride(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 && t.stride(1) == t.size(2),
              name, " must be [T, H, D] with dense inner dims");
  return t.stride(0);
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_BF16(x);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  const int rows = x.numel() / H;
  launch_rmsnorm(out.data_ptr(), x.data_ptr(), w.data_ptr(), (float)eps, rows,
                 H, cur_stream());
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual, torch::Tensor w,
                       double eps) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_CONTIG(residual);
  CHECK_BF16(x);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  const int rows = x.numel() / H;
  launch_fused_add_rmsnorm(x.data_ptr(), residual.data_ptr(), w.data_ptr(),
                           (float)eps, rows, H, cur_stream());
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_sin, bool is_neox) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat);
  const int T = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k.numel() > 0 ? k.size(1) : 0;
  const int rot = cos_sin.size(-1);
  const int64_t qs = row_stride(q, "q");
  const int64_t ks = Hk ? row_stride(k, "k") : 0;
  launch_rope(q.data_ptr(), Hk ? k.data_ptr() : nullptr,
              positions.data_ptr<int>(), cos_sin.data_ptr<float>(), T, Hq, Hk,
              D, rot, is_neox, qs, ks, cur_stream());
}

void rope_and_cache(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor k_cache, torch::Tensor v_cache,
                    torch::Tensor positions, torch::Tensor cos_sin,
                    torch::Tensor slot_mapping, bool is_neox, double k_scale,
                    double v_scale) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  const bool fp8 = is_fp8(k_cache);
  TORCH_CHECK(fp8 || k_cache.scalar_type() == at::kBFloat16,
              "k_cache must be bf16 or fp8_e4m3fn");
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k.size(1);
  const int BS = k_cache.size(2);
  const int rot = cos_sin.size(-1);
  TORCH_CHECK(rot == D || (rot < D && (D - rot) % 8 == 0));
  launch_rope_and_cache(
      q.data_ptr(), k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
      v_cache.data_ptr(), positions.data_ptr<int>(), cos_sin.data_ptr<float>(),
      slot_mapping.data_ptr<int64_t>(), T, Hq, Hk, D, rot, BS, is_neox, fp8,
      (float)k_scale, (float)v_scale, row_stride(q, "q"), row_stride(k, "k"),
      row_stride(v, "v"), cur_stream());
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor slot_mapping) {
  CHECK_GPU(k);
  CHECK_CONTIG(k);
  CHECK_BF16(k_cache);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T = k.size(0);
  const int Hk = k.size(1);
  const int D = k.size(2);
  const int BS = k_cache.size(2);
  TORCH_CHECK((Hk * D) % 8 == 0);
  launch_reshape_and_cache(k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
                           v_cache.data_ptr(), slot_mapping.data_ptr<int64_t>(),
                           T, Hk, D, BS, cur_stream());
}

void mla_reshape_and_cache(torch::Tensor latent, torch::Tensor k_rope,
                           torch::Tensor cache, torch::Tensor slot_mapping) {
  CHECK_GPU(latent);
  CHECK_CONTIG(latent);
  const int T = latent.size(0);
  const int R = latent.size(-1);
  const int DR = k_rope.size(-1);
  const int BS = cache.size(1);
  TORCH_CHECK(R % 8 == 0 && DR % 8 == 0);
  launch_mla_reshape_and_cache(latent.data_ptr(), k_rope.data_ptr(),
                               cache.data_ptr(),
                               slot_mapping.data_ptr<int64_t>(), T, R, DR, BS,
                               is_fp8(cache), 1.0f, cur_stream());
}

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_BF16(x);
  const int I = out.size(-1);
  TORCH_CHECK(I % 8 == 0);
  launch_act_and_mul(out.data_ptr(), x.data_ptr(), x.numel() / (2 * I), I,
                     false, cur_stream());
}

void gelu_and_mul(torch::Tensor out, torch::Tensor x) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_BF16(x);
  const int I = out.size(-1);
  TORCH_CHECK(I % 8 == 0);
  launch_act_and_mul(out.data_ptr(), x.data_ptr(), x.numel() / (2 * I), I, true,
                     cur_stream());
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor seq_lens,
                            double scale, int64_t sliding_window,
                            double softcap, torch::Tensor sinks,
                            int64_t max_seq_len, double k_scale,
                            double v_scale) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  const bool kv_fp8 = is_fp8(k_cache);
  TORCH_CHECK(kv_fp8 || k_cache.scalar_type() == at::kBFloat16,
              "k_cache must be bf16 or fp8_e4m3fn");
  TORCH_CHECK(block_tables.scalar_type() == at::kInt);
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  const int G = Hq / Hk;
  TORCH_CHECK(Hq % Hk == 0 && G <= 16, "GQA group must divide and be <= 16");
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  TORCH_CHECK(BS == 16 || BS == 32 || BS == 64, "block size must be 16/32/64");

  const float* sinks_ptr = nullptr;
  torch::Tensor sinks_f;
  if (sinks.numel() > 0) {
    sinks_f = sinks.to(at::kFloat).contiguous();
    sinks_ptr = sinks_f.data_ptr<float>();
  }

  // Flash-decoding split, graph-capture friendly (num_parts depends only on
  // max_seq_len and the batch bucket; partitions past a sequence's actual
  // length exit immediately / are skipped by the reduce kernel). Partition
  // size grows with B*Hk: when the batch alone fills the chip, bigger
  // partitions cut empty-partition sweep + reduce cost and pick the KT=128
  // tile path (A/B'd: profiles/README.md round 2).
  const int wg_base = B * Hk;
  int part_tokens = 256;
  if (wg_base >= 4096) part_tokens = 2048;
  else if (wg_base >= 1024) part_tokens = 1024;
  else if (wg_base >= 512) part_tokens = 512;
  int num_parts = (int)((max_seq_len + part_tokens - 1) / part_tokens);
  if (num_parts > 128) {
    part_tokens = (int)((max_seq_len + 127) / 128);
    part_tokens = (part_tokens + 127) / 128 * 128;  // chunk-aligned
    num_parts = (int)((max_seq_len + part_tokens - 1) / part_tokens);
  }
  torch::Tensor tmp_acc, tmp_ml;
  float *acc_ptr = nullptr, *ml_ptr = nullptr;
  if (num_parts > 1) {
    auto opts = q.options().dtype(at::kFloat);
    tmp_acc = torch::empty({B, Hq, num_parts, D}, opts);
    tmp_ml = torch::empty({B, Hq, num_parts, 2}, opts);
    acc_ptr = tmp_acc.data_ptr<float>();
    ml_ptr = tmp_ml.data_ptr<float>();
  }
  bool launched = false;
  static const bool use_valu = std::getenv("PARALLAX_ATTN_VALU") != nullptr;
  if (!use_valu || kv_fp8) {
    launch_paged_decode_mfma(
        out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), B, Hq, Hk, D,
        BS, max_blocks, row_stride(q, "q"), (float)scale, (int)sliding_window,
        (float)softcap, sinks_ptr, num_parts, part_tokens, acc_ptr, ml_ptr,
        kv_fp8, (float)k_scale, (float)v_scale, cur_stream(), &launched);
  }
  if (!launched) {
    launch_paged_attention_decode(
        out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), B, Hq, Hk, D,
        BS, max_blocks, row_stride(q, "q"), (float)scale, (int)sliding_window,
        (float)softcap, sinks_ptr, num_parts, part_tokens, acc_ptr, ml_ptr,
        cur_stream(), &launched);
  }
  TORCH_CHECK(launched, "no kernel instantiation for D=", D, " BS=", BS,
              " G=", G);
  if (num_parts > 1 && launched)
    launch_paged_attention_reduce(out.data_ptr(), acc_ptr, ml_ptr, B, Hq, D,
                                  num_parts, sinks_ptr, cur_stream());
}

void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_tables, torch::Tensor seq_lens,
                       torch::Tensor cu_q, torch::Tensor tile_req,
                       torch::Tensor tile_row0, double scale,
                       int64_t sliding_window, double softcap,
                       torch::Tensor sinks, double k_scale, double v_scale) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  const bool kv_fp8 = is_fp8(k_cache);
  TORCH_CHECK(kv_fp8 || k_cache.scalar_type() == at::kBFloat16,
              "k_cache must be bf16 or fp8_e4m3fn");
  TORCH_CHECK(block_tables.scalar_type() == at::kInt);
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  TORCH_CHECK(cu_q.scalar_type() == at::kInt);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  const int n_tiles = tile_req.size(0);
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");

  const float* sinks_ptr = nullptr;
  torch::Tensor sinks_f;
  if (sinks.numel() > 0) {
    sinks_f = sinks.to(at::kFloat).contiguous();
    sinks_ptr = sinks_f.data_ptr<float>();
  }
  bool launched = false;
  launch_prefill_attention(
      out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
      cu_q.data_ptr<int>(), tile_req.data_ptr<int>(), tile_row0.data_ptr<int>(),
      n_tiles, Hq, Hk, D, BS, max_blocks, row_stride(q, "q"), (float)scale,
      (int)sliding_window, (float)softcap, sinks_ptr, kv_fp8, (float)k_scale,
      (float)v_scale, cur_stream(), &launched);
  TORCH_CHECK(launched, "no prefill kernel instantiation for D=", D);
}

void mla_paged_attention_decode(torch::Tensor out, torch::Tensor q_latent,
                                torch::Tensor q_pe, torch::Tensor cache,
                                torch::Tensor block_tables,
                                torch::Tensor seq_lens, double scale,
                                int64_t max_seq_len, torch::Tensor topk_indices) {
  CHECK_GPU(q_latent);
  CHECK_CONTIG(q_latent);
  CHECK_CONTIG(q_pe);
  CHECK_BF16(q_latent);
  TORCH_CHECK(cache.scalar_type() == at::kBFloat16 || is_fp8(cache),
              "MLA cache must be bf16 or fp8_e4m3");
  TORCH_CHECK(block_tables.scalar_type() == at::kInt);
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  const int B = q_latent.size(0);
  const int H = q_latent.size(1);
  const int R = q_latent.size(2);
  const int DR = q_pe.size(2);
  const int BS = cache.size(1);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(cache.size(2) == R + DR, "cache entry dim mismatch");

  // same grid-fill-aware partition sizing as the GQA decode path
  const int mla_wg_base = B * ((H + 31) / 32);  // MLA_HBLOCK = 32
  int part_tokens = 256;
  if (mla_wg_base >= 4096) part_tokens = 2048;
  else if (mla_wg_base >= 1024) part_tokens = 1024;
  else if (mla_wg_base >= 512) part_tokens = 512;
  int num_parts = (int)((max_seq_len + part_tokens - 1) / part_tokens);
  if (num_parts > 128) {
    part_tokens = (int)((max_seq_len + 127) / 128);
    part_tokens = (part_tokens + 127) / 128 * 128;
    num_parts = (int)((max_seq_len + part_tokens - 1) / part_tokens);
  }
  torch::Tensor tmp_acc, tmp_ml;
  float *acc_ptr = nullptr, *ml_ptr = nullptr;
  if (num_parts > 1) {
    auto opts = q_latent.options().dtype(at::kFloat);
    tmp_acc = torch::empty({B, H, num_parts, R}, opts);
    tmp_ml = torch::empty({B, H, num_parts, 2}, opts);
    acc_ptr = tmp_acc.data_ptr<float>();
    ml_ptr = tmp_ml.data_ptr<float>();
  }
  const int* idx_ptr = nullptr;
  int max_topk = 0;
  if (topk_indices.numel() > 0) {
    TORCH_CHECK(topk_indices.scalar_type() == at::kInt &&
                topk_indices.is_contiguous());
    idx_ptr = topk_indices.data_ptr<int>();
    max_topk = topk_indices.size(1);
  }
  bool launched = false;
  launch_mla_paged_attention_decode(
      out.data_ptr(), q_latent.data_ptr(), q_pe.data_ptr(), cache.data_ptr(),
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), B, H, R, DR, BS,
      max_blocks, (float)scale, num_parts, part_tokens, acc_ptr, ml_ptr,
      idx_ptr, max_topk, is_fp8(cache), 1.0f, cur_stream(), &launched);
  TORCH_CHECK(launched, "no MLA kernel for R=", R, " DR=", DR, " BS=", BS);
  if (num_parts > 1)
    launch_paged_attention_reduce(out.data_ptr(), acc_ptr, ml_ptr, B, H, R,
                                  num_parts, nullptr, cur_stream());
}

void moe_forward(torch::Tensor out, torch::Tensor x, torch::Tensor w_gu,
                 torch::Tensor w_down, torch::Tensor perm,
                 torch::Tensor seg_offsets, torch::Tensor route_w,
                 int64_t topk, bool gelu, double limit,
                 torch::Tensor bias_gu) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_BF16(x);
  CHECK_BF16(w_gu);
  CHECK_BF16(w_down);
  TORCH_CHECK(out.scalar_type() == at::kFloat && out.is_contiguous());
  TORCH_CHECK(perm.scalar_type() == at::kLong);
  TORCH_CHECK(seg_offsets.scalar_type() == at::kInt);
  TORCH_CHECK(route_w.scalar_type() == at::kFloat);
  const int E = w_gu.size(0);
  const int I = w_gu.size(1) / 2;
  const int H = w_gu.size(2);
  const int n_assign = perm.size(0);
  TORCH_CHECK(H % 32 == 0 && I % 64 == 0,
              "MoE dims must be multiples of 32/64 (H=", H, " I=", I, ")");
  const int max_tiles = (n_assign + 15) / 16 + E;
  auto iopts = x.options().dtype(at::kInt);
  torch::Tensor tile_expert = torch::empty({max_tiles}, iopts);
  torch::Tensor tile_row0 = torch::empty({max_tiles}, iopts);
  torch::Tensor h_buf =
      torch::empty({n_assign, I}, x.options().dtype(at::kBFloat16));
  auto stream = cur_stream();
  launch_build_moe_tiles(tile_expert.data_ptr<int>(), tile_row0.data_ptr<int>(),
                         seg_offsets.data_ptr<int>(), E, max_tiles, stream);
  TORCH_CHECK(bias_gu.numel() == 0 ||
              (bias_gu.scalar_type() == at::kBFloat16 &&
               bias_gu.is_contiguous() && bias_gu.numel() == (int64_t)E * 2 * I));
  launch_moe_gate_up(h_buf.data_ptr(), x.data_ptr(), w_gu.data_ptr(),
                     tile_expert.data_ptr<int>(), tile_row0.data_ptr<int>(),
                     perm.data_ptr<int64_t>(), seg_offsets.data_ptr<int>(), E,
                     (int)topk, H, I, max_tiles, gelu, (float)limit,
                     bias_gu.numel() ? bias_gu.data_ptr() : nullptr, stream);
  launch_moe_down(out.data_ptr(), h_buf.data_ptr(), w_down.data_ptr(),
                  route_w.data_ptr<float>(), tile_expert.data_ptr<int>(),
                  tile_row0.data_ptr<int>(), perm.data_ptr<int64_t>(),
                  seg_offsets.data_ptr<int>(), E, (int)topk, H, I, max_tiles,
                  stream);
}

torch::Tensor dsa_indexer_scores(torch::Tensor q_index,
                                 torch::Tensor index_cache,
                                 torch::Tensor head_weights,
                                 torch::Tensor block_tables,
                                 torch::Tensor seq_lens, int64_t max_ctx) {
  CHECK_GPU(q_index);
  CHECK_BF16(q_index);
  CHECK_BF16(index_cache);
  TORCH_CHECK(index_cache.dim() == 3, "index cache must be [NB, BS, Di]");
  TORCH_CHECK(head_weights.scalar_type() == at::kFloat);
  const int B = q_index.size(0);
  const int Hi = q_index.size(1);
  const int Di = q_index.size(2);
  const int BS = index_cache.size(1);
  auto scores = torch::empty(
      {B, max_ctx}, q_index.options().dtype(at::kFloat));
  bool launched = false;
  launch_dsa_indexer_scores(
      scores.data_ptr<float>(), q_index.contiguous().data_ptr(),
      index_cache.data_ptr(), head_weights.contiguous().data_ptr<float>(),
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(), B, Hi, Di, BS,
      block_tables.size(1), (int)max_ctx, cur_stream(), &launched);
  TORCH_CHECK(launched, "no dsa_indexer kernel for Hi=", Hi, " Di=", Di);
  return scores;
}

void store_indexer_cache(torch::Tensor index_keys, torch::Tensor index_cache,
                         torch::Tensor slot_mapping, int64_t trash_slot) {
  CHECK_GPU(index_keys);
  CHECK_BF16(index_keys);
  CHECK_BF16(index_cache);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T = index_keys.size(0);
  const int DI = index_keys.size(-1);
  TORCH_CHECK(DI % 8 == 0);
  launch_store_indexer_cache(index_cache.data_ptr(),
                             index_keys.contiguous().data_ptr(),
                             slot_mapping.data_ptr<int64_t>(), T, trash_slot,
                             DI, cur_stream());
}

torch::Tensor msa_block_scores(torch::Tensor q, torch::Tensor k_cache,
                               torch::Tensor block_tables,
                               torch::Tensor seq_lens, int64_t sparse_block,
                               int64_t max_sparse_blocks) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  CHECK_BF16(k_cache);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  auto out = torch::empty(
      {B, max_sparse_blocks}, q.options().dtype(at::kFloat));
  bool launched = false;
  launch_msa_block_scores(out.data_ptr<float>(), q.contiguous().data_ptr(),
                          k_cache.data_ptr(), block_tables.data_ptr<int>(),
                          seq_lens.data_ptr<int>(), B, Hq, Hk, D, BS,
                          block_tables.size(1), (int)sparse_block,
                          (int)max_sparse_blocks, cur_stream(), &launched);
  TORCH_CHECK(launched, "no msa_block_scores kernel for D=", D);
  return out;
}

torch::Tensor msa_topk_tokens(torch::Tensor block_scores,
                              torch::Tensor seq_lens, int64_t sparse_block,
                              int64_t topk_blocks, int64_t init_blocks,
                              int64_t local_blocks, int64_t max_positions) {
  CHECK_GPU(block_scores);
  TORCH_CHECK(block_scores.scalar_type() == at::kFloat);
  const int B = block_scores.size(0);
  auto out = torch::empty(
      {B, max_positions}, block_scores.options().dtype(at::kLong));
  launch_msa_topk_tokens(out.data_ptr<int64_t>(),
                         block_scores.contiguous().data_ptr<float>(),
                         seq_lens.data_ptr<int>(), B, (int)sparse_block,
                         block_scores.size(1), (int)topk_blocks,
                         (int)init_blocks, (int)local_blocks,
                         (int)max_positions, cur_stream());
  return out;
}

void moe_forward_fp8(torch::Tensor out, torch::Tensor x, torch::Tensor w_gu,
                     torch::Tensor w_gu_scale, torch::Tensor w_down,
                     torch::Tensor w_down_scale, torch::Tensor perm,
                     torch::Tensor seg_offsets, torch::Tensor route_w,
                     int64_t topk, bool gelu, double limit) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_BF16(x);
  TORCH_CHECK(w_gu.scalar_type() == at::kFloat8_e4m3fn && w_gu.is_contiguous());
  TORCH_CHECK(w_down.scalar_type() == at::kFloat8_e4m3fn &&
              w_down.is_contiguous());
  TORCH_CHECK(w_gu_scale.scalar_type() == at::kFloat &&
              w_down_scale.scalar_type() == at::kFloat);
  TORCH_CHECK(out.scalar_type() == at::kFloat && out.is_contiguous());
  TORCH_CHECK(perm.scalar_type() == at::kLong);
  TORCH_CHECK(seg_offsets.scalar_type() == at::kInt);
  TORCH_CHECK(route_w.scalar_type() == at::kFloat);
  const int T = x.size(0);
  const int E = w_gu.size(0);
  const int I = w_gu.size(1) / 2;
  const int H = w_gu.size(2);
  const int n_assign = perm.size(0);
  TORCH_CHECK(H % 32 == 0 && I % 64 == 0,
              "MoE dims must be multiples of 32/64 (H=", H, " I=", I, ")");
  const int max_tiles = (n_assign + 15) / 16 + E;
  auto iopts = x.options().dtype(at::kInt);
  auto fopts = x.options().dtype(at::kFloat);
  auto bopts = x.options().dtype(at::kByte);
  torch::Tensor tile_expert = torch::empty({max_tiles}, iopts);
  torch::Tensor tile_row0 = torch::empty({max_tiles}, iopts);
  torch::Tensor x_fp8 = torch::empty({T, H}, bopts);
  torch::Tensor x_scale = torch::empty({T}, fopts);
  torch::Tensor h_buf =
      torch::empty({n_assign, I}, x.options().dtype(at::kBFloat16));
  torch::Tensor h_fp8 = torch::empty({n_assign, I}, bopts);
  torch::Tensor h_scale = torch::empty({std::max(n_assign, 1)}, fopts);
  auto stream = cur_stream();
  launch_quantize_fp8_rows(x_fp8.data_ptr(), x_scale.data_ptr<float>(),
                           x.data_ptr(), T, H, stream);
  launch_build_moe_tiles(tile_expert.data_ptr<int>(), tile_row0.data_ptr<int>(),
                         seg_offsets.data_ptr<int>(), E, max_tiles, stream);
  launch_moe_gate_up_fp8(
      h_buf.data_ptr(), x_fp8.data_ptr(), x_scale.data_ptr<float>(),
      w_gu.data_ptr(), w_gu_scale.data_ptr<float>(),
      tile_expert.data_ptr<int>(), tile_row0.data_ptr<int>(),
      perm.data_ptr<int64_t>(), seg_offsets.data_ptr<int>(), E, (int)topk, H,
      I, max_tiles, gelu, (float)limit, stream);
  if (n_assign > 0)
    launch_quantize_fp8_rows(h_fp8.data_ptr(), h_scale.data_ptr<float>(),
                             h_buf.data_ptr(), n_assign, I, stream);
  launch_moe_down_fp8(
      out.data_ptr(), h_fp8.data_ptr(), h_scale.data_ptr<float>(),
      w_down.data_ptr(), w_down_scale.data_ptr<float>(),
      route_w.data_ptr<float>(), tile_expert.data_ptr<int>(),
      tile_row0.data_ptr<int>(), perm.data_ptr<int64_t>(),
      seg_offsets.data_ptr<int>(), E, (int)topk, H, I, max_tiles, stream);
}

torch::Tensor sample_gumbel(torch::Tensor logits, torch::Tensor inv_temp,
                            torch::Tensor greedy, int64_t seed) {
  CHECK_GPU(logits);
  CHECK_CONTIG(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat, "logits must be fp32");
  TORCH_CHECK(inv_temp.scalar_type() == at::kFloat);
  TORCH_CHECK(greedy.scalar_type() == at::kByte);
  const int B = logits.size(0);
  const int V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(at::kLong));
  launch_sample_gumbel(out.data_ptr<int64_t>(), logits.data_ptr<float>(),
                       inv_temp.contiguous().data_ptr<float>(),
                       greedy.contiguous().data_ptr<uint8_t>(),
                       (uint64_t)seed, B, V, cur_stream());
  return out;
}

bool skinny_gemm(torch::Tensor c, torch::Tensor x, torch::Tensor w,
                 torch::Tensor bias) {
  CHECK_GPU(x);
  CHECK_BF16(x);
  CHECK_BF16(w);
  TORCH_CHECK(w.is_contiguous());
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1);
  TORCH_CHECK(c.dim() == 2 && c.stride(1) == 1);
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  bool launched = false;
  launch_skinny_gemm(c.data_ptr(), x.data_ptr(), w.data_ptr(),
                     bias.numel() ? bias.data_ptr() : nullptr, M, N, K,
                     x.stride(0), c.stride(0), cur_stream(), &launched);
  return launched;
}

void msa_paged_attention_decode(torch::Tensor out, torch::Tensor q,
                                torch::Tensor k_cache, torch::Tensor v_cache,
                                torch::Tensor block_tables,
                                torch::Tensor seq_lens,
                                torch::Tensor token_positions, double scale) {
  CHECK_GPU(q);
  CHECK_BF16(q);
  CHECK_BF16(k_cache);
  TORCH_CHECK(token_positions.scalar_type() == at::kInt &&
              token_positions.is_contiguous());
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  // token_positions: [B, P] shared across kv heads, or [B, Hk, P] per head
  const int pos_heads = token_positions.dim() == 3 ? token_positions.size(1) : 1;
  if (pos_heads > 1)
    TORCH_CHECK(pos_heads == Hk, "positions head dim must equal Hk");
  bool launched = false;
  launch_msa_paged_attention_decode(
      out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
      token_positions.data_ptr<int>(), token_positions.size(-1), pos_heads, B,
      Hq, Hk, D, BS, block_tables.size(1), row_stride(q, "q"), (float)scale,
      cur_stream(), &launched);
  TORCH_CHECK(launched, "no MSA kernel for D=", D, " BS=", BS, " G=", Hq / Hk);
}

torch::Tensor lt_linear(torch::Tensor x, torch::Tensor w);  // lt_gemm.cpp
torch::Tensor lt_linear_fp8(torch::Tensor x_q, torch::Tensor w_q,
                            torch::Tensor x_scale, torch::Tensor w_scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("lt_linear_fp8", &lt_linear_fp8,
        "fp8-E4M3 W8A8 y = (xq*sx) @ (wq*sw)^T (bf16 out)");
  m.def("lt_linear", &lt_linear,
        "tuned hipBLASLt y = x @ w^T (bf16, fp32 accum)");
  m.def("msa_paged_attention_decode", &msa_paged_attention_decode);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("sample_gumbel", &sample_gumbel);
  m.def("moe_forward", &moe_forward);
  m.def("moe_forward_fp8", &moe_forward_fp8);
  m.def("dsa_indexer_scores", &dsa_indexer_scores);
  m.def("store_indexer_cache", &store_indexer_cache);
  m.def("msa_block_scores", &msa_block_scores);
  m.def("msa_topk_tokens", &msa_topk_tokens);
  m.def("mla_paged_attention_decode", &mla_paged_attention_decode);
  m.def("prefill_attention", &prefill_attention);
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("rope_inplace", &rope_inplace);
  m.def("rope_and_cache", &rope_and_cache);
  m.def("reshape_and_cache", &reshape_and_cache);
  m.def("mla_reshape_and_cache", &mla_reshape_and_cache);
  m.def("silu_and_mul", &silu_and_mul);
  m.def("gelu_and_mul", &gelu_and_mul);
  m.def("paged_attention_decode", &paged_attention_decode);
}
