// Python bindings for the sutro-amd CDNA4 kernels.
// Tensors are validated here; raw pointers + the current HIP stream go to the
// extern "C" launchers in the .hip translation units.
#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#define CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) \
  TORCH_CHECK((x).scalar_type() == at::kBFloat16, #x " must be bf16")

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

extern "C" {
void sutro_rmsnorm(void*, const void*, const void*, float, long, int,
                   hipStream_t);
void sutro_fused_add_rmsnorm(void*, void*, const void*, float, long, int,
                             hipStream_t);
void sutro_silu_mul(void*, const void*, long, int, hipStream_t);
void sutro_rope_and_cache(void*, void*, const void*, const long*, const long*,
                          void*, void*, const float*, int, int, int, int, int,
                          hipStream_t);
static bool is_fp8(const torch::Tensor& t) {
  return t.scalar_type() == at::kFloat8_e4m3fn;
}
void sutro_mean_pool_normalize(float*, const void*, const int*, int, int,
                               hipStream_t);
void sutro_attn_decode(void*, const void*, const void*, const void*,
                       const int*, const int*, int, int, int, int, int, int,
                       int, float, hipStream_t);
void sutro_attn_prefill(void*, const void*, const void*, const void*,
                        const int*, const int*, const int*, const int*,
                        const int*, int, int, int, int, int, int, float,
                        hipStream_t);
void sutro_mfma32_probe(float*, const void*, const void*, hipStream_t);
void sutro_gemm_tn_launch(void*, const void*, const void*, const void*, int,
                          int, long, int, int, int, int, hipStream_t);
void sutro_mfma16_probe(float*, const void*, const void*, hipStream_t);
void sutro_hd64_stage_probe(float*, float*, void*, const void*, const void*,
                            const void*, int, float, hipStream_t);
void sutro_qkv_prep(const void*, void*, void*, void*, const long*, const long*,
                    const float*, const void*, const void*, float, int, int,
                    int, int, int, int, int, hipStream_t);
void sutro_sampler_fused(const void*, int, const float*, const float*,
                         const int*, const float*, const unsigned int*, int,
                         long, int, int, int*, float*, hipStream_t);
void sutro_grouped_gemm(void*, const void*, const void*, const int*,
                        const int*, const int*, int, int, int, long, int,
                        int, hipStream_t);
void sutro_moe_combine(void*, const void*, const long*, const float*, int,
                       int, int, hipStream_t);
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x); CHECK_BF16(x);
  const long C = x.size(-1);
  const long rows = x.numel() / C;
  TORCH_CHECK(C % 8 == 0 && C <= 16384, "unsupported row size ", C);
  sutro_rmsnorm(out.data_ptr(), x.data_ptr(), w.data_ptr(), (float)eps, rows,
                (int)C, cur_stream());
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual, torch::Tensor w,
                       double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x); CHECK_BF16(x); CHECK_CONTIG(residual);
  const long C = x.size(-1);
  const long rows = x.numel() / C;
  TORCH_CHECK(C % 8 == 0 && C <= 16384, "unsupported row size ", C);
  sutro_fused_add_rmsnorm(x.data_ptr(), residual.data_ptr(), w.data_ptr(),
                          (float)eps, rows, (int)C, cur_stream());
}

void silu_mul(torch::Tensor out, torch::Tensor x) {
  CHECK_CUDA(x); CHECK_CONTIG(x); CHECK_BF16(x);
  const long I = x.size(-1) / 2;
  const long T = x.numel() / (2 * I);
  TORCH_CHECK(I % 8 == 0, "intermediate size must be a multiple of 8");
  sutro_silu_mul(out.data_ptr(), x.data_ptr(), T, (int)I, cur_stream());
}

void rope_and_cache(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor positions, torch::Tensor slot_mapping,
                    torch::Tensor k_cache, torch::Tensor v_cache,
                    torch::Tensor cos_sin) {
  CHECK_CUDA(q); CHECK_CONTIG(q); CHECK_BF16(q);
  CHECK_CONTIG(k); CHECK_CONTIG(v); CHECK_CONTIG(k_cache);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat, "cos_sin must be f32");
  const int T = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hk = k.size(1);
  const int bs = k_cache.size(2);
  TORCH_CHECK(D % 2 == 0 && D / 2 <= 128, "unsupported head_dim ", D);
  sutro_rope_and_cache(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       positions.data_ptr<long>(), slot_mapping.data_ptr<long>(),
                       k_cache.data_ptr(), v_cache.data_ptr(),
                       cos_sin.data_ptr<float>(), T, Hq, Hk, D, bs,
                       cur_stream());
}

void mean_pool_normalize(torch::Tensor out, torch::Tensor hidden,
                         torch::Tensor qlocs) {
  CHECK_CUDA(hidden); CHECK_CONTIG(hidden); CHECK_BF16(hidden);
  const int S = out.size(0), H = hidden.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  sutro_mean_pool_normalize(out.data_ptr<float>(), hidden.data_ptr(),
                            qlocs.data_ptr<int>(), S, H, cur_stream());
}

void paged_attention(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                     torch::Tensor v_cache, torch::Tensor block_tables,
                     torch::Tensor seq_lens, torch::Tensor qlocs, double scale,
                     int64_t num_decodes, torch::Tensor tile_seq,
                     torch::Tensor tile_q0, int64_t prefill_token_count) {
  CHECK_CUDA(q); CHECK_CONTIG(q); CHECK_BF16(q); CHECK_CONTIG(k_cache);
  const int Hq = q.size(1), D = q.size(2);
  const int Hk = k_cache.size(1);
  const int bs = k_cache.size(2);
  const int S = seq_lens.size(0);
  TORCH_CHECK(D == 128 || D == 64,
              "attention kernels support head_dim 64/128 (got ", D, ")");
  TORCH_CHECK(bs == 32, "attention kernels require kv_block_size 32");
  TORCH_CHECK(Hq % Hk == 0 && Hq / Hk <= 8,
              "GQA group size must divide and be <= 8");
  const int bt_stride = block_tables.size(1);
  const int n_tiles = tile_seq.numel();
  if (n_tiles > 0) {
    sutro_attn_prefill(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                       v_cache.data_ptr(), block_tables.data_ptr<int>(),
                       seq_lens.data_ptr<int>(), qlocs.data_ptr<int>(),
                       tile_seq.data_ptr<int>(), tile_q0.data_ptr<int>(),
                       n_tiles, bt_stride, Hq, Hk, D,
                       is_fp8(k_cache) ? 1 : 0, (float)scale, cur_stream());
  }
  if (num_decodes > 0) {
    const long dec_off = prefill_token_count;  // decode rows are the tail
    const int seq_offset = S - (int)num_decodes;
    const u_int16_t* qp = (const u_int16_t*)q.data_ptr();
    u_int16_t* op = (u_int16_t*)out.data_ptr();
    sutro_attn_decode(op + dec_off * Hq * D, qp + dec_off * Hq * D,
                      k_cache.data_ptr(), v_cache.data_ptr(),
                      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
                      bt_stride, (int)num_decodes, Hq, Hk, D,
                      is_fp8(k_cache) ? 1 : 0, seq_offset, (float)scale,
                      cur_stream());
  }
}

void qkv_prep(torch::Tensor qkv, torch::Tensor q_out, torch::Tensor k_cache,
              torch::Tensor v_cache, torch::Tensor positions,
              torch::Tensor slot_mapping, torch::Tensor cos_sin,
              c10::optional<torch::Tensor> q_norm_w,
              c10::optional<torch::Tensor> k_norm_w, double eps) {
  CHECK_CUDA(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
  CHECK_CONTIG(q_out); CHECK_CONTIG(k_cache); CHECK_CONTIG(v_cache);
  const int T = qkv.size(0);
  const int Hq = q_out.size(1), D = q_out.size(2);
  const int Hk = k_cache.size(1);
  const int bs = k_cache.size(2);
  const int row_stride = qkv.size(1);
  TORCH_CHECK(row_stride == Hq * D + 2 * Hk * D, "qkv width mismatch");
  const void* qw = q_norm_w ? q_norm_w->data_ptr() : nullptr;
  const void* kw = k_norm_w ? k_norm_w->data_ptr() : nullptr;
  sutro_qkv_prep(qkv.data_ptr(), q_out.data_ptr(), k_cache.data_ptr(),
                 v_cache.data_ptr(), positions.data_ptr<long>(),
                 slot_mapping.data_ptr<long>(), cos_sin.data_ptr<float>(), qw,
                 kw, (float)eps, T, Hq, Hk, D, bs, row_stride,
                 is_fp8(k_cache) ? 1 : 0, cur_stream());
}

// out[M,N] = x[M,K] @ w[N,K]^T (+ res), bf16 in/out, fp32 accumulate.
// bm/bn pick the macro-tile; M%bm==0 and K%64==0 required (caller falls back
// to torch.mm otherwise). res, when given, is added to the output (fused
// residual epilogue).
torch::Tensor gemm_tn(torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> res, long bm, long bn,
                      long swz, long xcd_swz) {
  CHECK_CUDA(x); CHECK_BF16(x); CHECK_CUDA(w); CHECK_BF16(w);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous(), "contiguous required");
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(M % bm == 0 && K % 64 == 0, "shape not tile-divisible");
  auto out = torch::empty({M, N}, x.options());
  const void* rp = nullptr;
  if (res.has_value()) {
    CHECK_CUDA(*res); CHECK_BF16(*res);
    TORCH_CHECK(res->is_contiguous() && res->size(0) == M && res->size(1) == N,
                "res shape");
    rp = res->data_ptr();
  }
  sutro_gemm_tn_launch(out.data_ptr(), x.data_ptr(), w.data_ptr(), rp, (int)M,
                       (int)N, K, (int)bm, (int)bn, (int)swz, (int)xcd_swz,
                       cur_stream());
  return out;
}

torch::Tensor mfma32_probe(torch::Tensor a, torch::Tensor b) {
  CHECK_CUDA(a); CHECK_BF16(a);
  auto c = torch::zeros({32, 32}, a.options().dtype(at::kFloat));
  sutro_mfma32_probe(c.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                     cur_stream());
  return c;
}

std::vector<torch::Tensor> hd64_stage_probe(torch::Tensor q, torch::Tensor k,
                                             torch::Tensor v, long L,
                                             double scale) {
  CHECK_CUDA(q); CHECK_BF16(q);
  auto fopt = q.options().dtype(at::kFloat);
  auto vt = torch::zeros({64 * 40}, fopt);
  auto p = torch::zeros({16 * 40}, fopt);
  auto out = torch::zeros({1, 64}, q.options());
  sutro_hd64_stage_probe(vt.data_ptr<float>(), p.data_ptr<float>(),
                         out.data_ptr(), q.data_ptr(), k.data_ptr(),
                         v.data_ptr(), (int)L, (float)scale, cur_stream());
  return {vt, p, out};
}

torch::Tensor mfma16_probe(torch::Tensor a, torch::Tensor b) {
  CHECK_CUDA(a); CHECK_BF16(a);
  auto c = torch::zeros({16, 16}, a.options().dtype(at::kFloat));
  sutro_mfma16_probe(c.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                     cur_stream());
  return c;
}

void grouped_gemm(torch::Tensor out, torch::Tensor a, torch::Tensor w,
                  c10::optional<torch::Tensor> row_tok, torch::Tensor tile_off,
                  torch::Tensor counts, long max_tiles, bool gate_silu,
                  long bm) {
  TORCH_CHECK(bm == 64 || bm == 128, "bm must be 64 or 128");
  CHECK_CUDA(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_CUDA(w); CHECK_CONTIG(w); CHECK_BF16(w);
  CHECK_CONTIG(out); CHECK_BF16(out);
  const long E = w.size(0), N = w.size(1), K = w.size(2);
  const long n_cols = gate_silu ? N / 2 : N;
  TORCH_CHECK(K % 64 == 0 && n_cols % 64 == 0,
              "grouped_gemm needs K, n_cols multiples of 64 (K=", K,
              ", n_cols=", n_cols, ")");
  TORCH_CHECK(a.size(1) == K, "A K mismatch");
  TORCH_CHECK(out.size(1) == n_cols, "out width mismatch");
  TORCH_CHECK(tile_off.scalar_type() == at::kInt
              && counts.scalar_type() == at::kInt, "tile_off/counts int32");
  TORCH_CHECK(tile_off.numel() == E + 1 && counts.numel() == E, "seg sizes");
  const int* rt = nullptr;
  if (row_tok.has_value()) {
    TORCH_CHECK(row_tok->scalar_type() == at::kInt, "row_tok int32");
    rt = row_tok->data_ptr<int>();
  }
  sutro_grouped_gemm(out.data_ptr(), a.data_ptr(), w.data_ptr(), rt,
                     tile_off.data_ptr<int>(), counts.data_ptr<int>(), (int)E,
                     (int)max_tiles, (int)n_cols, K, gate_silu ? 1 : 0,
                     (int)bm, cur_stream());
}

void moe_combine(torch::Tensor out, torch::Tensor rows, torch::Tensor padpos,
                 torch::Tensor w) {
  CHECK_CUDA(rows); CHECK_CONTIG(rows); CHECK_BF16(rows);
  CHECK_CONTIG(out); CHECK_BF16(out);
  const long T = out.size(0), h = out.size(1), k = padpos.size(1);
  TORCH_CHECK(h % 8 == 0, "h must be a multiple of 8");
  TORCH_CHECK(padpos.scalar_type() == at::kLong, "padpos must be int64");
  TORCH_CHECK(w.scalar_type() == at::kFloat, "w must be f32");
  TORCH_CHECK(padpos.is_contiguous() && w.is_contiguous(), "contig");
  sutro_moe_combine(out.data_ptr(), rows.data_ptr(),
                    padpos.data_ptr<long>(), w.data_ptr<float>(), (int)T,
                    (int)h, (int)k, cur_stream());
}

void sampler_fused(torch::Tensor logits, torch::Tensor temps,
                   torch::Tensor topps, torch::Tensor topks, torch::Tensor us,
                   c10::optional<torch::Tensor> mask, long vl,
                   torch::Tensor out_tok, torch::Tensor out_lp) {
  CHECK_CUDA(logits); CHECK_CONTIG(logits);
  const bool f32 = logits.scalar_type() == at::kFloat;
  TORCH_CHECK(f32 || logits.scalar_type() == at::kBFloat16,
              "logits must be bf16 or f32");
  const long n = logits.size(0), v_row = logits.size(1);
  TORCH_CHECK(vl <= v_row, "vocab_limit exceeds logits row");
  TORCH_CHECK(temps.numel() >= n && topps.numel() >= n && topks.numel() >= n
              && us.numel() >= n, "param tensors too small");
  TORCH_CHECK(topks.scalar_type() == at::kInt, "topks must be int32");
  const unsigned int* mptr = nullptr;
  int w_words = 0;
  if (mask.has_value()) {
    auto& mt = mask.value();
    CHECK_CUDA(mt); CHECK_CONTIG(mt);
    TORCH_CHECK(mt.scalar_type() == at::kInt, "mask must be int32 packed");
    TORCH_CHECK(mt.size(0) == n, "mask rows mismatch");
    w_words = (int)mt.size(1);
    TORCH_CHECK((long)w_words * 32 >= vl, "mask too narrow for vocab_limit");
    mptr = (const unsigned int*)mt.data_ptr();
  }
  sutro_sampler_fused(logits.data_ptr(), f32 ? 1 : 0,
                      temps.data_ptr<float>(), topps.data_ptr<float>(),
                      topks.data_ptr<int>(), us.data_ptr<float>(), mptr,
                      (int)n, v_row, (int)vl, w_words,
                      out_tok.data_ptr<int>(), out_lp.data_ptr<float>(),
                      cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm, "residual+=x; x=rmsnorm");
  m.def("silu_mul", &silu_mul, "fused SwiGLU");
  m.def("rope_and_cache", &rope_and_cache, "RoPE + paged KV write");
  m.def("qkv_prep", &qkv_prep, "fused qk-norm + RoPE + KV write + q gather");
  m.def("mean_pool_normalize", &mean_pool_normalize, "varlen mean pool + L2");
  m.def("paged_attention", &paged_attention, "paged prefill+decode attention");
  m.def("mfma32_probe", &mfma32_probe, "MFMA fragment-layout probe");
  m.def("mfma16_probe", &mfma16_probe, "16x16 MFMA fragment-layout probe");
  m.def("hd64_stage_probe", &hd64_stage_probe, "D=64 decode stage dump probe");
  m.def("moe_combine", &moe_combine, "fused MoE weighted gather-combine");
  m.def("grouped_gemm", &grouped_gemm,
        "dropless MoE grouped GEMM (padded segments, static grid)");
  m.def("sampler_fused", &sampler_fused,
        "fused mask+temp+topk/topp+sample+logprob (radix descent, no sort)");
  m.def("gemm_tn", &gemm_tn, "bf16 TN GEMM (MFMA, glds dbuf)",
        py::arg("x"), py::arg("w"), py::arg("res") = c10::nullopt,
        py::arg("bm") = 256, py::arg("bn") = 256, py::arg("swz") = 2,
        py::arg("xcd_swz") = 0);
}
