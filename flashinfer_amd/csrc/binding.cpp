// PyTorch bindings for the MI355X kernel library. Single torch-header TU —
// all device code lives in the .hip TUs and is reached via extern "C"
// launchers (fast incremental builds: kernel TUs compile in seconds).
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "fi/params.hpp"
namespace fi_ext = fi;

extern "C" {
hipError_t fi_norm_quant(int add, int dtype, void* x, void* residual, const void* w,
                         void* out, const float* scale, int rows, int d, float eps,
                         hipStream_t stream);
hipError_t fi_layernorm_quant(int dtype, const void* x, const void* w, const void* b,
                              void* out, const float* scale, int rows, int d,
                              float eps, hipStream_t stream);
hipError_t fi_norm(int which, int dtype, const void* x, const void* w, const void* b,
                   void* y, void* residual, int rows, int d, int64_t sx, int64_t sy,
                   float eps, int weight_bias, hipStream_t stream);
hipError_t fi_act_and_mul(int which, int dtype, const void* in, void* out, int64_t tokens,
                          int d, hipStream_t stream);
hipError_t fi_rope(int dtype, fi_ext::RopeParams* p, hipStream_t stream);
hipError_t fi_append_paged_kv_cache(int dtype, int cache_dtype, void* k_data,
                                    void* v_data, const int32_t* indices,
                                    const int32_t* indptr,
                                    const int32_t* last_page_len, int page_size,
                                    int num_heads, int head_dim, int64_t stride_page,
                                    int64_t stride_n, int64_t stride_h, const void* k,
                                    const void* v, const int32_t* batch_indices,
                                    const int32_t* positions, int64_t nnz,
                                    int64_t k_stride_n, int64_t k_stride_h,
                                    int64_t v_stride_n, int64_t v_stride_h,
                                    float k_scale, float v_scale, hipStream_t stream);
hipError_t fi_batch_indices_positions(const int32_t* append_indptr, const int32_t* seq_lens,
                                      int32_t* batch_indices, int32_t* positions, int batch,
                                      hipStream_t stream);
hipError_t fi_merge_states(int in_dtype, int out_dtype, const void* v_in, const float* s_in,
                           void* v_out, float* s_out, const int32_t* merge_indptr,
                           int64_t uniform_count, int64_t num_pos, int num_heads,
                           int head_dim, hipStream_t stream);
hipError_t fi_merge_state_in_place(int dtype, void* v, float* s, const void* v_other,
                                   const float* s_other, int64_t num_pos, int num_heads,
                                   int head_dim, const uint8_t* mask, hipStream_t stream);
hipError_t fi_batch_decode(int dtype, int kv_dtype, fi_ext::DecodeParams* p,
                           hipStream_t stream);
hipError_t fi_batch_decode_fused(int dtype, int kv_dtype, fi_ext::DecodeParams* p,
                                 hipStream_t stream);
hipError_t fi_decode_mfma(int dtype, int kv_dtype, fi_ext::DecodeParams* p,
                          hipStream_t stream);
namespace fi_ext2 {
struct HolisticParams {
  fi_ext::PrefillParams pf;
  fi_ext::DecodeParams dec;
  const int32_t* items;
  int n_pf_items;
  int n_items;
  int n_dec_wgs;
  uint32_t* queue_head;
};
}  // namespace fi_ext2
hipError_t fi_batch_attention(int dtype, fi_ext2::HolisticParams* h, int group_dec,
                              int causal, int n_wgs, hipStream_t stream);
hipError_t fi_gemm_nt(int dtype, const void* A, const void* B, void* C, int M, int N,
                      int K, int64_t lda, int64_t ldb, int64_t ldc, float alpha,
                      hipStream_t stream);
hipError_t fi_gemm_nt_v2(const void* A, const void* B, void* C, int M, int N, int K,
                         int64_t lda, int64_t ldb, int64_t ldc, float alpha,
                         hipStream_t stream);
hipError_t fi_batch_prefill(int dtype, int kv_dtype, fi_ext::PrefillParams* p,
                            int paged, hipStream_t stream);
hipError_t fi_softmax(fi_ext::SamplingParams* p, hipStream_t stream);
hipError_t fi_sampling(int mode, int from_logits, fi_ext::SamplingParams* p,
                       hipStream_t stream);
hipError_t fi_renorm(int which, fi_ext::SamplingParams* p, hipStream_t stream);
hipError_t fi_chain_speculative(fi_ext::SpecParams* sp, hipStream_t stream);
hipError_t fi_mla_decode(int dtype, int kv_dtype, fi_ext::MlaParams* p,
                         hipStream_t stream);
hipError_t fi_group_gemm_nt(const void* A, const void* W, void* C,
                            const int32_t* m_indptr, const int32_t* w_indices,
                            int num_segments, int max_m_tiles, int N, int K,
                            int64_t lda, int64_t ldw_n, int64_t ldw_seg, int64_t ldc,
                            hipStream_t stream);
hipError_t fi_gemm_fp8_grouped(const void* A, const void* W, void* C,
                               const int32_t* m_indptr, const int32_t* w_indices,
                               int num_segments, int max_m_tiles, int N, int K,
                               int64_t lda, int64_t ldw_n, int64_t ldw_seg, int64_t ldc,
                               const float* a_scales, const float* b_scales,
                               float scalar_scale, int64_t a_scale_stride,
                               int flat_tiles, int mx, hipStream_t stream);
hipError_t fi_per_group_quant_fp8(int dtype, int trans_scale, const void* x, void* q,
                                  float* scale, int64_t rows, int K, int64_t stride_row,
                                  float eps, hipStream_t stream);
hipError_t fi_scale_quant_fp8(int dtype, const void* x, void* q, const float* inv_scale,
                              int64_t n, hipStream_t stream);
hipError_t fi_topk(const float* x, float* out_v, int32_t* out_i,
                   const int32_t* lengths, const int32_t* row_starts,
                   const int32_t* offsets, const int32_t* page_table,
                   const int32_t* row_to_batch, const int32_t* page_table_row_starts,
                   int pt_cols, int page_size, int mode, int rows, int d, int k,
                   int64_t stride, int tie_break, hipStream_t stream);
hipError_t fi_packbits(const uint8_t* x, uint8_t* y, int64_t n, hipStream_t stream);
hipError_t fi_segment_packbits(const uint8_t* x, uint8_t* y, const int32_t* x_indptr,
                               const int32_t* y_indptr, int num_segments,
                               hipStream_t stream);
hipError_t fi_selective_state_update(int dtype, fi_ext::SSUParams* p,
                                     hipStream_t stream);
hipError_t fi_gather_rows(int dtype, const void* src, void* dst, const int32_t* row_map,
                          int64_t rows, int cols, hipStream_t stream);
hipError_t fi_moe_topk_softmax(const float* logits, float* weights, int32_t* ids,
                               int T, int E, int k, int renorm, hipStream_t stream);
hipError_t fi_dsv3_routing(const float* logits, const float* bias, float* weights,
                           int32_t* ids, int T, int E, int k, int n_group,
                           int topk_group, float scale, hipStream_t stream);
hipError_t fi_moe_build_permute(const int32_t* ids, int32_t* counts,
                                int32_t* m_indptr, int32_t* cursor,
                                int32_t* token_of_copy, int32_t* inv, int n,
                                int k, int E, int align, hipStream_t stream);
hipError_t fi_gather_quant(int dtype, const void* src, const int32_t* token_of_copy,
                           uint8_t* dst, float* scale, int R, int K, int e8m0,
                           hipStream_t stream);
hipError_t fi_silu_mul_quant(int dtype, const void* h, uint8_t* dst, float* scale,
                             int R, int I, int gelu, int e8m0, hipStream_t stream);
hipError_t fi_moe_finalize(int dtype, const void* h, void* out, const int32_t* pos,
                           const float* w, int64_t tokens, int topk, int cols,
                           hipStream_t stream);
hipError_t fi_gdn_decode(int dtype, int state_f32, int per_channel_gate, void* state,
                         const void* q, const void* k, const void* v, const float* g,
                         const float* beta, void* out, int B, int H, int Dk, int Dv,
                         hipStream_t stream);
hipError_t fi_one_shot_ar(int dtype, fi::ARParams* p, void* out, int64_t numel,
                          hipStream_t stream);
hipError_t fi_two_shot_ar(int dtype, fi_ext::ARParams* p, unsigned long long shard_off,
                          void* out, int64_t numel, hipStream_t stream);
hipError_t fi_one_shot_ar_rmsnorm(int dtype, fi::ARParams* p, void* out,
                                  void* residual, const void* weight, int rows,
                                  int d, float eps, hipStream_t stream);
hipError_t fi_qk_rope(int dtype, const void* qkv, const void* qw, const void* kw,
                      const float* cos_t, const float* sin_t, void* q_out,
                      void* k_out, void* v_out, int64_t tokens, int S, int Hq,
                      int Hk, int Hv, int D, float eps, float attn_factor,
                      int qk_norm, hipStream_t stream);
hipError_t fi_mhc_post(int dtype, const void* x, const void* residual,
                       const float* post_mix, const float* comb_mix, void* out,
                       int64_t tokens, int H, hipStream_t stream);
hipError_t fi_mhc_pre(int dtype, int prenorm, const float* dot_mix,
                      const float* sqrsum, const void* residual, const float* scale,
                      const float* base, float* post_mix, float* comb_mix,
                      void* layer_input, int64_t tokens, int H, int num_splits,
                      float inv_k, float rms_eps, float pre_eps, float sink_eps,
                      float post_mult, int sink_repeat, hipStream_t stream);
hipError_t fi_concat_mla_k(int dtype, void* k, const void* k_nope, const void* k_rope,
                           int64_t tokens, int Hk, int nope, int rope,
                           hipStream_t stream);
hipError_t fi_ssd_scan(int dtype, const void* x, const float* dt, const float* A,
                       const void* Bm, const void* Cm, const float* D, const void* z,
                       const float* dt_bias, const float* init_states,
                       float* final_states, void* out, int batch, int L, int H, int G,
                       int P, int dstate, int dt_softplus, float dt_min, float dt_max,
                       int d_has_hdim, hipStream_t stream);
hipError_t fi_gdn_chunk(int dtype, int per_channel_gate, const void* q, const void* k,
                        const void* v, const float* gate, const float* beta, void* out,
                        const int32_t* cu_seqlens, const float* init_state,
                        float* final_state, float scale, int num_seqs, int H, int D,
                        hipStream_t stream);
}

namespace {

int dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kBFloat16: return 0;
    case at::kHalf: return 1;
    case at::kFloat: return 2;
    case at::ScalarType::Float8_e4m3fn: return 3;
    case at::kByte: return 3;  // fp8 cache passed as uint8 view
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.device().index()).stream();
}

void check_hip(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, what, " failed: ", hipGetErrorString(e));
}


// ---------------- norm ----------------

void norm_common(int which, at::Tensor x, at::Tensor w, c10::optional<at::Tensor> b,
                 c10::optional<at::Tensor> y, c10::optional<at::Tensor> residual,
                 double eps, bool weight_bias) {
  TORCH_CHECK(x.is_cuda(), "input must be on GPU");
  TORCH_CHECK(x.dim() == 2, "expect 2D [rows, hidden]");
  TORCH_CHECK(x.stride(1) == 1, "innermost dim must be contiguous");
  int rows = x.size(0), d = x.size(1);
  const void* bp = b.has_value() ? b->data_ptr() : nullptr;
  void* yp = y.has_value() ? y->data_ptr() : nullptr;
  void* rp = residual.has_value() ? residual->data_ptr() : nullptr;
  int64_t sy = y.has_value() ? y->stride(0) : (residual.has_value() ? residual->stride(0) : 0);
  check_hip(fi_norm(which, dtype_code(x), x.data_ptr(), w.data_ptr(), bp, yp, rp, rows, d,
                    x.stride(0), sy, (float)eps, weight_bias, cur_stream(x)),
            "fi_norm");
}

void rmsnorm(at::Tensor x, at::Tensor w, at::Tensor out, double eps, bool weight_bias) {
  norm_common(0, x, w, c10::nullopt, out, c10::nullopt, eps, weight_bias);
}

void fused_add_rmsnorm(at::Tensor x, at::Tensor residual, at::Tensor w, double eps,
                       bool weight_bias) {
  norm_common(1, x, w, c10::nullopt, c10::nullopt, residual, eps, weight_bias);
}

void layernorm(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> b, at::Tensor out,
               double eps) {
  norm_common(2, x, w, b, out, c10::nullopt, eps, false);
}

void rmsnorm_silu(at::Tensor x, at::Tensor w, at::Tensor out, double eps) {
  norm_common(3, x, w, c10::nullopt, out, c10::nullopt, eps, false);
}

void layernorm_quant(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> b,
                     at::Tensor out, at::Tensor scale, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(out.scalar_type() == at::kFloat8_e4m3fn && out.is_contiguous());
  check_hip(fi_layernorm_quant(dtype_code(x), x.data_ptr(), w.data_ptr(),
                               b ? b->data_ptr() : nullptr, out.data_ptr(),
                               scale.data_ptr<float>(), x.size(0), x.size(1),
                               (float)eps, cur_stream(x)),
            "fi_layernorm_quant");
}

void rmsnorm_quant(at::Tensor x, c10::optional<at::Tensor> residual, at::Tensor w,
                   at::Tensor out, at::Tensor scale, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(out.scalar_type() == at::kFloat8_e4m3fn && out.is_contiguous());
  TORCH_CHECK(scale.scalar_type() == at::kFloat && scale.is_cuda());
  check_hip(fi_norm_quant(residual.has_value(), dtype_code(x), x.data_ptr(),
                          residual ? residual->data_ptr() : nullptr, w.data_ptr(),
                          out.data_ptr(), scale.data_ptr<float>(), x.size(0),
                          x.size(1), (float)eps, cur_stream(x)),
            "fi_norm_quant");
}

// ---------------- activation ----------------

void act_and_mul(int which, at::Tensor x, at::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  int d = x.size(-1) / 2;
  int64_t tokens = x.numel() / (2 * d);
  check_hip(fi_act_and_mul(which, dtype_code(x), x.data_ptr(), out.data_ptr(), tokens, d,
                           cur_stream(x)),
            "fi_act_and_mul");
}

// ---------------- rope ----------------

void apply_rope(at::Tensor q, at::Tensor k, at::Tensor q_out, at::Tensor k_out,
                at::Tensor pos_ids, c10::optional<at::Tensor> cos_sin_cache, int64_t rot_dim,
                bool interleave, double rope_scale, double rope_theta, double smooth_a,
                double smooth_b, double rcp_factor) {
  TORCH_CHECK(q.is_cuda() && k.is_cuda());
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [nnz, heads, head_dim]");
  TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1);
  TORCH_CHECK(pos_ids.scalar_type() == at::kInt);
  fi_ext::RopeParams p{};
  p.q = q.data_ptr();
  p.k = k.data_ptr();
  p.q_out = q_out.data_ptr();
  p.k_out = k_out.data_ptr();
  p.pos_ids = pos_ids.data_ptr<int32_t>();
  p.cos_sin_cache = cos_sin_cache.has_value() ? cos_sin_cache->data_ptr<float>() : nullptr;
  p.nnz = q.size(0);
  p.num_qo_heads = q.size(1);
  p.num_kv_heads = k.size(1);
  p.head_dim = q.size(2);
  p.rot_dim = (int)rot_dim;
  p.q_stride_n = q.stride(0);
  p.q_stride_h = q.stride(1);
  p.k_stride_n = k.stride(0);
  p.k_stride_h = k.stride(1);
  p.o_q_stride_n = q_out.stride(0);
  p.o_q_stride_h = q_out.stride(1);
  p.o_k_stride_n = k_out.stride(0);
  p.o_k_stride_h = k_out.stride(1);
  p.rope_rcp_scale = (float)(1.0 / rope_scale);
  p.rope_theta = (float)rope_theta;
  p.smooth_a = (float)smooth_a;
  p.smooth_b = (float)smooth_b;
  p.rcp_factor = (float)rcp_factor;
  p.interleave = interleave;
  check_hip(fi_rope(dtype_code(q), &p, cur_stream(q)), "fi_rope");
}

// ---------------- page ----------------

void append_paged_kv_cache(at::Tensor k, at::Tensor v, at::Tensor batch_indices,
                           at::Tensor positions, at::Tensor k_cache, at::Tensor v_cache,
                           at::Tensor indices, at::Tensor indptr, at::Tensor last_page_len,
                           int64_t layout /*0 NHD, 1 HND*/, double k_scale,
                           double v_scale) {
  TORCH_CHECK(k.is_cuda() && k.dim() == 3);
  // k_cache: NHD [pages, page_size, H, D] or HND [pages, H, page_size, D]
  int page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
  int num_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
  int head_dim = k_cache.size(3);
  int64_t stride_page = k_cache.stride(0);
  int64_t stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
  int64_t stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
  check_hip(fi_append_paged_kv_cache(
                dtype_code(k), dtype_code(k_cache), k_cache.data_ptr(),
                v_cache.data_ptr(), indices.data_ptr<int32_t>(),
                indptr.data_ptr<int32_t>(), last_page_len.data_ptr<int32_t>(), page_size,
                num_heads, head_dim, stride_page, stride_n, stride_h, k.data_ptr(),
                v.data_ptr(), batch_indices.data_ptr<int32_t>(),
                positions.data_ptr<int32_t>(), k.size(0), k.stride(0), k.stride(1),
                v.stride(0), v.stride(1), (float)k_scale, (float)v_scale,
                cur_stream(k)),
            "fi_append_paged_kv_cache");
}

void get_batch_indices_positions(at::Tensor append_indptr, at::Tensor seq_lens,
                                 at::Tensor batch_indices, at::Tensor positions) {
  int batch = append_indptr.size(0) - 1;
  check_hip(fi_batch_indices_positions(append_indptr.data_ptr<int32_t>(),
                                       seq_lens.data_ptr<int32_t>(),
                                       batch_indices.data_ptr<int32_t>(),
                                       positions.data_ptr<int32_t>(), batch,
                                       cur_stream(append_indptr)),
            "fi_batch_indices_positions");
}

// ---------------- merge ----------------

void merge_states(at::Tensor v_in, at::Tensor s_in, at::Tensor v_out,
                  c10::optional<at::Tensor> s_out, c10::optional<at::Tensor> merge_indptr,
                  int64_t uniform_count, int64_t num_pos) {
  int num_heads = v_out.size(-2);
  int head_dim = v_out.size(-1);
  check_hip(fi_merge_states(dtype_code(v_in), dtype_code(v_out), v_in.data_ptr(),
                            s_in.data_ptr<float>(), v_out.data_ptr(),
                            s_out.has_value() ? s_out->data_ptr<float>() : nullptr,
                            merge_indptr.has_value() ? merge_indptr->data_ptr<int32_t>()
                                                     : nullptr,
                            uniform_count, num_pos, num_heads, head_dim, cur_stream(v_in)),
            "fi_merge_states");
}

void merge_state_in_place(at::Tensor v, at::Tensor s, at::Tensor v_other,
                          at::Tensor s_other, c10::optional<at::Tensor> mask) {
  int num_heads = v.size(-2), head_dim = v.size(-1);
  int64_t num_pos = v.numel() / (num_heads * head_dim);
  check_hip(fi_merge_state_in_place(dtype_code(v), v.data_ptr(), s.data_ptr<float>(),
                                    v_other.data_ptr(), s_other.data_ptr<float>(), num_pos,
                                    num_heads, head_dim,
                                    mask.has_value() ? mask->data_ptr<uint8_t>() : nullptr,
                                    cur_stream(v)),
            "fi_merge_state_in_place");
}

// ---------------- decode ----------------

void batch_decode_run(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                      at::Tensor kv_indices, at::Tensor kv_indptr,
                      at::Tensor kv_last_page_len, int64_t layout, at::Tensor work_req,
                      at::Tensor work_chunk, int64_t chunk_size, at::Tensor tmp_v,
                      at::Tensor tmp_s, double sm_scale, double logits_soft_cap,
                      int64_t window_left, bool alibi) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "q must be [batch, num_qo_heads, head_dim]");
  fi_ext::DecodeParams p{};
  p.q = q.data_ptr();
  p.k_data = k_cache.data_ptr();
  p.v_data = v_cache.data_ptr();
  p.kv_indices = kv_indices.data_ptr<int32_t>();
  p.kv_indptr = kv_indptr.data_ptr<int32_t>();
  p.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  int page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
  p.page_size = fi::uint_fastdiv((uint32_t)page_size);
  p.num_kv_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
  p.num_qo_heads = q.size(1);
  p.head_dim = q.size(2);
  p.stride_page = k_cache.stride(0);
  p.stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
  p.stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
  p.work_req = work_req.data_ptr<int32_t>();
  p.work_chunk = work_chunk.data_ptr<int32_t>();
  p.n_items = work_req.size(0);
  p.chunk_size = (int)chunk_size;
  p.tmp_v = (float*)tmp_v.data_ptr();  // stored in the q dtype (see mla.py)
  p.tmp_s = tmp_s.data_ptr<float>();
  p.q_stride_n = q.stride(0);
  p.q_stride_h = q.stride(1);
  p.sm_scale = (float)sm_scale;
  p.logits_soft_cap = (float)logits_soft_cap;
  p.window_left = (int)window_left;
  p.alibi = alibi ? 1 : 0;
  check_hip(fi_batch_decode(dtype_code(q), dtype_code(k_cache), &p, cur_stream(q)),
            "fi_batch_decode");
}

// Fused whole-request decode (one workgroup per (req, kv_head), in-LDS merge,
// direct output) — the short-kv/small-batch latency path.
void batch_decode_fused_run(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                            at::Tensor kv_indices, at::Tensor kv_indptr,
                            at::Tensor kv_last_page_len, int64_t layout,
                            at::Tensor out, c10::optional<at::Tensor> lse,
                            double sm_scale, double logits_soft_cap,
                            int64_t window_left, bool alibi) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "q must be [batch, num_qo_heads, head_dim]");
  fi_ext::DecodeParams p{};
  p.q = q.data_ptr();
  p.k_data = k_cache.data_ptr();
  p.v_data = v_cache.data_ptr();
  p.kv_indices = kv_indices.data_ptr<int32_t>();
  p.kv_indptr = kv_indptr.data_ptr<int32_t>();
  p.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  int page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
  p.page_size = fi::uint_fastdiv((uint32_t)page_size);
  p.num_kv_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
  p.num_qo_heads = q.size(1);
  p.head_dim = q.size(2);
  p.stride_page = k_cache.stride(0);
  p.stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
  p.stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
  p.q_stride_n = q.stride(0);
  p.q_stride_h = q.stride(1);
  p.sm_scale = (float)sm_scale;
  p.logits_soft_cap = (float)logits_soft_cap;
  p.window_left = (int)window_left;
  p.alibi = alibi ? 1 : 0;
  p.o = out.data_ptr();
  p.lse = lse.has_value() ? lse->data_ptr<float>() : nullptr;
  p.o_stride_n = out.stride(0);
  p.o_stride_h = out.stride(1);
  p.batch = q.size(0);
  check_hip(fi_batch_decode_fused(dtype_code(q), dtype_code(k_cache), &p, cur_stream(q)),
            "fi_batch_decode_fused");
}

// MFMA fused decode (GQA group >= 8): same argument surface as the fused
// vector run, different kernel family (csrc/attention/decode_mfma.hip).
void batch_decode_mfma_run(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                           at::Tensor kv_indices, at::Tensor kv_indptr,
                           at::Tensor kv_last_page_len, int64_t layout,
                           at::Tensor out, c10::optional<at::Tensor> lse,
                           double sm_scale, double logits_soft_cap,
                           int64_t window_left, bool alibi, int64_t split,
                           c10::optional<at::Tensor> tmp_v,
                           c10::optional<at::Tensor> tmp_s,
                           c10::optional<at::Tensor> counters) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "q must be [batch, num_qo_heads, head_dim]");
  TORCH_CHECK(q.stride(2) == 1 && out.stride(2) == 1);
  fi_ext::DecodeParams p{};
  p.q = q.data_ptr();
  p.k_data = k_cache.data_ptr();
  p.v_data = v_cache.data_ptr();
  p.kv_indices = kv_indices.data_ptr<int32_t>();
  p.kv_indptr = kv_indptr.data_ptr<int32_t>();
  p.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  int page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
  p.page_size = fi::uint_fastdiv((uint32_t)page_size);
  p.num_kv_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
  p.num_qo_heads = q.size(1);
  p.head_dim = q.size(2);
  p.stride_page = k_cache.stride(0);
  p.stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
  p.stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
  p.q_stride_n = q.stride(0);
  p.q_stride_h = q.stride(1);
  p.sm_scale = (float)sm_scale;
  p.logits_soft_cap = (float)logits_soft_cap;
  p.window_left = (int)window_left;
  p.alibi = alibi ? 1 : 0;
  p.o = out.data_ptr();
  p.lse = lse.has_value() ? lse->data_ptr<float>() : nullptr;
  p.o_stride_n = out.stride(0);
  p.o_stride_h = out.stride(1);
  p.batch = q.size(0);
  p.split = (int)split;
  if (split > 1) {
    TORCH_CHECK(tmp_v.has_value() && tmp_s.has_value(),
                "split mfma decode needs tmp_v/tmp_s");
    p.tmp_v = (float*)tmp_v->data_ptr();
    p.tmp_s = tmp_s->data_ptr<float>();
    // same-XCD in-kernel merge (host gates on (batch*kv_heads)%8==0)
    p.counters = counters.has_value()
                     ? (uint32_t*)counters->data_ptr<int32_t>()
                     : nullptr;
  }
  check_hip(fi_decode_mfma(dtype_code(q), dtype_code(k_cache), &p,
                           cur_stream(q)),
            "fi_decode_mfma");
}

// Persistent holistic BatchAttention: one launch over a tagged work queue of
// prefill tiles + decode items (csrc/attention/batch_attention.hip).
void batch_attention_run(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                         at::Tensor qo_indptr, at::Tensor kv_indices,
                         at::Tensor kv_indptr, at::Tensor kv_last_page_len,
                         int64_t layout, at::Tensor items, at::Tensor queue_head,
                         at::Tensor out, c10::optional<at::Tensor> lse,
                         double sm_scale, double logits_soft_cap,
                         int64_t window_left, bool causal, int64_t group_dec,
                         int64_t n_wgs, int64_t n_pf_items, int64_t n_dec_wgs) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "q must be [nnz, Hq, D]");
  TORCH_CHECK(q.stride(2) == 1 && out.stride(2) == 1);
  fi_ext2::HolisticParams h{};
  // prefill side
  fi_ext::PrefillParams& pf = h.pf;
  pf.q = q.data_ptr();
  pf.out = out.data_ptr();
  pf.lse = lse.has_value() ? lse->data_ptr<float>() : nullptr;
  pf.qo_indptr = qo_indptr.data_ptr<int32_t>();
  pf.k_data = k_cache.data_ptr();
  pf.v_data = v_cache.data_ptr();
  pf.kv_indices = kv_indices.data_ptr<int32_t>();
  pf.kv_indptr = kv_indptr.data_ptr<int32_t>();
  pf.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  int page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
  int num_kv_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
  pf.page_size = fi::uint_fastdiv((uint32_t)page_size);
  pf.kv_stride_page = k_cache.stride(0);
  pf.kv_stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
  pf.kv_stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
  pf.v_stride_page = v_cache.stride(0);
  pf.v_stride_n = layout == 0 ? v_cache.stride(1) : v_cache.stride(2);
  pf.v_stride_h = layout == 0 ? v_cache.stride(2) : v_cache.stride(1);
  pf.head_dim_vo = v_cache.size(3);
  pf.num_qo_heads = q.size(1);
  pf.num_kv_heads = num_kv_heads;
  pf.head_dim = q.size(2);
  pf.group = fi::uint_fastdiv((uint32_t)(pf.num_qo_heads / num_kv_heads));
  pf.q_stride_n = q.stride(0);
  pf.q_stride_h = q.stride(1);
  pf.o_stride_n = out.stride(0);
  pf.o_stride_h = out.stride(1);
  pf.sm_scale = (float)sm_scale;
  pf.logits_soft_cap = (float)logits_soft_cap;
  pf.window_left = (int)window_left;
  pf.causal = causal ? 1 : 0;
  pf.cta_q = 256;
  // decode side (shares the page table; q rows indexed by qo_indptr[req])
  fi_ext::DecodeParams& d = h.dec;
  d.k_data = k_cache.data_ptr();
  d.v_data = v_cache.data_ptr();
  d.kv_indices = kv_indices.data_ptr<int32_t>();
  d.kv_indptr = kv_indptr.data_ptr<int32_t>();
  d.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  d.page_size = fi::uint_fastdiv((uint32_t)page_size);
  d.num_kv_heads = num_kv_heads;
  d.num_qo_heads = q.size(1);
  d.head_dim = q.size(2);
  d.stride_page = pf.kv_stride_page;
  d.stride_n = pf.kv_stride_n;
  d.stride_h = pf.kv_stride_h;
  d.q = q.data_ptr();
  d.q_stride_n = q.stride(0);
  d.q_stride_h = q.stride(1);
  d.o = out.data_ptr();
  d.lse = pf.lse;
  d.o_stride_n = out.stride(0);
  d.o_stride_h = out.stride(1);
  d.sm_scale = (float)sm_scale;
  d.logits_soft_cap = (float)logits_soft_cap;
  d.window_left = (int)window_left;
  d.split = 1;
  h.items = items.data_ptr<int32_t>();
  h.n_items = items.size(0);
  h.n_pf_items = (int)n_pf_items;
  h.n_dec_wgs = (int)n_dec_wgs;
  h.queue_head = (uint32_t*)queue_head.data_ptr();
  check_hip(fi_batch_attention(dtype_code(q), &h, (int)group_dec, causal ? 1 : 0,
                               (int)n_wgs, cur_stream(q)),
            "fi_batch_attention");
}

// ---------------- prefill ----------------

// paged: k_cache/v_cache are 4-D cache planes, kv_indptr is a page range.
// ragged: k_cache/v_cache are [nnz_kv, Hkv, D], kv_indptr is token offsets.
void batch_prefill_run(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                       at::Tensor qo_indptr, c10::optional<at::Tensor> kv_indices,
                       at::Tensor kv_indptr, c10::optional<at::Tensor> kv_last_page_len,
                       int64_t layout, at::Tensor tile_req, at::Tensor tile_qstart,
                       at::Tensor out, c10::optional<at::Tensor> lse, double sm_scale,
                       double logits_soft_cap, int64_t window_left, bool causal,
                       bool paged, int64_t cta_q,
                       c10::optional<at::Tensor> mask_data,
                       c10::optional<at::Tensor> mask_byte_indptr, bool alibi,
                       double k_descale, double v_descale,
                       c10::optional<at::Tensor> prof_buf = c10::nullopt,
                       c10::optional<at::Tensor> tile_kv_chunk = c10::nullopt,
                       int64_t kv_chunk = 0,
                       c10::optional<at::Tensor> req_slot_base = c10::nullopt,
                       c10::optional<at::Tensor> tmp_v = c10::nullopt,
                       c10::optional<at::Tensor> tmp_s = c10::nullopt,
                       int64_t custom_fn = 0,
                       c10::optional<at::Tensor> tile_req_b = c10::nullopt,
                       c10::optional<at::Tensor> tile_qstart_b = c10::nullopt) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3, "q must be [nnz, Hq, D]");
  TORCH_CHECK(q.stride(2) == 1 && out.stride(2) == 1);
  fi_ext::PrefillParams p{};
  p.q = q.data_ptr();
  p.out = out.data_ptr();
  p.lse = lse.has_value() ? lse->data_ptr<float>() : nullptr;
  p.qo_indptr = qo_indptr.data_ptr<int32_t>();
  p.k_data = k_cache.data_ptr();
  p.v_data = v_cache.data_ptr();
  p.kv_indices = kv_indices.has_value() ? kv_indices->data_ptr<int32_t>() : nullptr;
  p.kv_indptr = kv_indptr.data_ptr<int32_t>();
  p.kv_last_page_len =
      kv_last_page_len.has_value() ? kv_last_page_len->data_ptr<int32_t>() : nullptr;
  int num_kv_heads, page_size;
  if (paged) {
    TORCH_CHECK(k_cache.dim() == 4);
    page_size = layout == 0 ? k_cache.size(1) : k_cache.size(2);
    num_kv_heads = layout == 0 ? k_cache.size(2) : k_cache.size(1);
    p.kv_stride_page = k_cache.stride(0);
    p.kv_stride_n = layout == 0 ? k_cache.stride(1) : k_cache.stride(2);
    p.kv_stride_h = layout == 0 ? k_cache.stride(2) : k_cache.stride(1);
    p.v_stride_page = v_cache.stride(0);
    p.v_stride_n = layout == 0 ? v_cache.stride(1) : v_cache.stride(2);
    p.v_stride_h = layout == 0 ? v_cache.stride(2) : v_cache.stride(1);
    p.head_dim_vo = v_cache.size(3);
  } else {
    TORCH_CHECK(k_cache.dim() == 3);
    page_size = 1;
    num_kv_heads = k_cache.size(1);
    p.kv_stride_page = 0;
    p.kv_stride_n = k_cache.stride(0);
    p.kv_stride_h = k_cache.stride(1);
    p.v_stride_page = 0;
    p.v_stride_n = v_cache.stride(0);
    p.v_stride_h = v_cache.stride(1);
    p.head_dim_vo = v_cache.size(2);
  }
  p.page_size = fi::uint_fastdiv((uint32_t)page_size);
  p.tile_req = tile_req.data_ptr<int32_t>();
  p.tile_qstart = tile_qstart.data_ptr<int32_t>();
  p.tile_req_b = tile_req_b.has_value() ? tile_req_b->data_ptr<int32_t>()
                                        : nullptr;
  p.tile_qstart_b = tile_qstart_b.has_value()
                        ? tile_qstart_b->data_ptr<int32_t>()
                        : nullptr;
  p.n_tiles = tile_req.size(0);
  p.num_qo_heads = q.size(1);
  p.num_kv_heads = num_kv_heads;
  p.head_dim = q.size(2);
  TORCH_CHECK(p.num_qo_heads % num_kv_heads == 0);
  p.group = fi::uint_fastdiv((uint32_t)(p.num_qo_heads / num_kv_heads));
  p.q_stride_n = q.stride(0);
  p.q_stride_h = q.stride(1);
  p.o_stride_n = out.stride(0);
  p.o_stride_h = out.stride(1);
  p.sm_scale = (float)sm_scale;
  p.logits_soft_cap = (float)logits_soft_cap;
  p.window_left = (int)window_left;
  p.causal = causal ? 1 : 0;
  p.cta_q = (int)cta_q;
  p.mask_data = mask_data.has_value() ? mask_data->data_ptr<uint8_t>() : nullptr;
  p.mask_byte_indptr =
      mask_byte_indptr.has_value() ? mask_byte_indptr->data_ptr<int32_t>() : nullptr;
  p.alibi = alibi ? 1 : 0;
  p.k_descale = (float)k_descale;
  p.v_descale = (float)v_descale;
  if (prof_buf.has_value()) {
    TORCH_CHECK(prof_buf->scalar_type() == at::kUInt64 && prof_buf->is_cuda());
    p.prof_buf = (unsigned long long*)prof_buf->data_ptr();
  }
  if (tile_kv_chunk.has_value()) {
    TORCH_CHECK(req_slot_base.has_value() && tmp_v.has_value() && tmp_s.has_value(),
                "split-KV prefill needs req_slot_base/tmp_v/tmp_s");
    p.tile_kv_chunk = tile_kv_chunk->data_ptr<int32_t>();
    p.kv_chunk = (int)kv_chunk;
    p.req_slot_base = req_slot_base->data_ptr<int32_t>();
    p.tmp_v = (float*)tmp_v->data_ptr();
    p.tmp_s = tmp_s->data_ptr<float>();
  }
  if (custom_fn) {
    // JIT attention variant: dispatch through the custom module's entry
    // (same signature as fi_batch_prefill; fn resolved by ctypes in
    // flashinfer_amd/jit/attention.py)
    auto fn = (hipError_t (*)(int, fi_ext::PrefillParams*, int, hipStream_t))
        (void*)custom_fn;
    check_hip(fn(dtype_code(q), &p, paged ? 1 : 0, cur_stream(q)),
              "fi_batch_prefill_custom");
    return;
  }
  check_hip(fi_batch_prefill(dtype_code(q), dtype_code(k_cache), &p, paged ? 1 : 0,
                             cur_stream(q)),
            "fi_batch_prefill");
}

// ---------------- sampling ----------------

fi_ext::SamplingParams sampling_params_common(at::Tensor probs,
                                              c10::optional<at::Tensor> out_probs,
                                              c10::optional<at::Tensor> out_ids,
                                              c10::optional<at::Tensor> uniforms,
                                              c10::optional<at::Tensor> top_k,
                                              c10::optional<at::Tensor> top_p,
                                              c10::optional<at::Tensor> row_indices,
                                              double scalar_p, int64_t scalar_k,
                                              double min_p) {
  TORCH_CHECK(probs.is_cuda() && probs.dim() == 2 && probs.stride(1) == 1);
  TORCH_CHECK(probs.scalar_type() == at::kFloat, "sampling ops take f32");
  fi_ext::SamplingParams p{};
  p.probs = probs.data_ptr<float>();
  p.out_probs = out_probs.has_value() ? out_probs->data_ptr<float>() : nullptr;
  p.out_ids = out_ids.has_value() ? out_ids->data_ptr<int32_t>() : nullptr;
  p.uniforms = uniforms.has_value() ? uniforms->data_ptr<float>() : nullptr;
  p.top_p = top_p.has_value() ? top_p->data_ptr<float>() : nullptr;
  p.top_k = top_k.has_value() ? top_k->data_ptr<int32_t>() : nullptr;
  p.row_indices = row_indices.has_value() ? row_indices->data_ptr<int32_t>() : nullptr;
  p.scalar_p = (float)scalar_p;
  p.scalar_k = (int)scalar_k;
  p.min_p = (float)min_p;
  p.rows = row_indices.has_value() ? row_indices->size(0)
           : (out_ids.has_value() ? out_ids->size(0) : probs.size(0));
  p.vocab = probs.size(1);
  p.rounds = uniforms.has_value() && uniforms->dim() == 2 ? uniforms->size(1) : 1;
  p.stride_row = probs.stride(0);
  p.temperature = 1.f;
  return p;
}

void softmax_op(at::Tensor logits, at::Tensor out, double temperature) {
  auto p = sampling_params_common(logits, out, c10::nullopt, c10::nullopt,
                                  c10::nullopt, c10::nullopt, c10::nullopt, 0, 0, 0);
  p.temperature = (float)temperature;
  check_hip(fi_softmax(&p, cur_stream(logits)), "fi_softmax");
}

void sampling_op(int64_t mode, bool from_logits, at::Tensor probs, at::Tensor out_ids,
                 at::Tensor uniforms, c10::optional<at::Tensor> top_k,
                 c10::optional<at::Tensor> top_p, c10::optional<at::Tensor> row_indices,
                 double scalar_p, int64_t scalar_k, double min_p) {
  auto p = sampling_params_common(probs, c10::nullopt, out_ids, uniforms, top_k, top_p,
                                  row_indices, scalar_p, scalar_k, min_p);
  check_hip(fi_sampling((int)mode, from_logits ? 1 : 0, &p, cur_stream(probs)),
            "fi_sampling");
}

void renorm_op(int64_t which, at::Tensor probs, at::Tensor out,
               c10::optional<at::Tensor> top_k, c10::optional<at::Tensor> top_p,
               double scalar_p, int64_t scalar_k) {
  auto p = sampling_params_common(probs, out, c10::nullopt, c10::nullopt, top_k, top_p,
                                  c10::nullopt, scalar_p, scalar_k, 0);
  p.rows = probs.size(0);
  check_hip(fi_renorm((int)which, &p, cur_stream(probs)), "fi_renorm");
}

void chain_speculative_op(at::Tensor draft_probs, at::Tensor draft_ids,
                          at::Tensor target_probs, at::Tensor out_ids,
                          at::Tensor accepted_num, at::Tensor emitted_num,
                          at::Tensor uniforms) {
  fi_ext::SpecParams sp{};
  sp.draft_probs = draft_probs.data_ptr<float>();
  sp.draft_ids = draft_ids.data_ptr<int32_t>();
  sp.target_probs = target_probs.data_ptr<float>();
  sp.out_ids = out_ids.data_ptr<int32_t>();
  sp.accepted_num = accepted_num.data_ptr<int32_t>();
  sp.emitted_num = emitted_num.data_ptr<int32_t>();
  sp.uniforms = uniforms.data_ptr<float>();
  sp.B = draft_probs.size(0);
  sp.n = draft_probs.size(1);
  sp.vocab = draft_probs.size(2);
  check_hip(fi_chain_speculative(&sp, cur_stream(draft_probs)), "fi_chain_speculative");
}

// ---------------- mla ----------------

void mla_run(at::Tensor q_nope, at::Tensor q_pe, at::Tensor ckv_cache,
             at::Tensor kpe_cache, at::Tensor qo_indptr, at::Tensor kv_indices,
             at::Tensor kv_indptr, at::Tensor kv_last_page_len, at::Tensor tile_req,
             at::Tensor tile_row0, at::Tensor tile_chunk, int64_t chunk_size,
             int64_t max_chunks, at::Tensor tmp_v, at::Tensor tmp_s, double sm_scale,
             bool causal, double ckv_descale, double kpe_descale) {
  TORCH_CHECK(q_nope.is_cuda() && q_nope.dim() == 3 && q_nope.size(2) == 512);
  TORCH_CHECK(q_pe.size(2) == 64);
  TORCH_CHECK(ckv_cache.dim() == 3 && ckv_cache.size(2) == 512,
              "ckv cache must be [pages, page_size, 512]");
  fi_ext::MlaParams p{};
  p.q_nope = q_nope.data_ptr();
  p.q_pe = q_pe.data_ptr();
  p.ckv_data = ckv_cache.data_ptr();
  p.kpe_data = kpe_cache.data_ptr();
  p.qo_indptr = qo_indptr.data_ptr<int32_t>();
  p.kv_indices = kv_indices.data_ptr<int32_t>();
  p.kv_indptr = kv_indptr.data_ptr<int32_t>();
  p.kv_last_page_len = kv_last_page_len.data_ptr<int32_t>();
  p.page_size = fi::uint_fastdiv((uint32_t)ckv_cache.size(1));
  p.num_heads_fd = fi::uint_fastdiv((uint32_t)q_nope.size(1));
  p.ckv_stride_page = ckv_cache.stride(0);
  p.ckv_stride_n = ckv_cache.stride(1);
  p.kpe_stride_page = kpe_cache.stride(0);
  p.kpe_stride_n = kpe_cache.stride(1);
  p.q_nope_stride_n = q_nope.stride(0);
  p.q_nope_stride_h = q_nope.stride(1);
  p.q_pe_stride_n = q_pe.stride(0);
  p.q_pe_stride_h = q_pe.stride(1);
  p.tile_req = tile_req.data_ptr<int32_t>();
  p.tile_row0 = tile_row0.data_ptr<int32_t>();
  p.tile_chunk = tile_chunk.data_ptr<int32_t>();
  p.n_items = tile_req.size(0);
  p.chunk_size = (int)chunk_size;
  p.max_chunks = (int)max_chunks;
  p.num_heads = q_nope.size(1);
  p.tmp_v = (float*)tmp_v.data_ptr();  // stored in the q dtype (see mla.py)
  p.tmp_s = tmp_s.data_ptr<float>();
  p.sm_scale = (float)sm_scale;
  p.causal = causal ? 1 : 0;
  p.ckv_descale = (float)ckv_descale;
  p.kpe_descale = (float)kpe_descale;
  check_hip(fi_mla_decode(dtype_code(q_nope), dtype_code(ckv_cache), &p,
                          cur_stream(q_nope)),
            "fi_mla_decode");
}

// ---------------- gemm ----------------

// C[M,N] = A[M,K] @ B_nt[N,K]^T ; all K-contiguous row-major.
void gemm_nt(at::Tensor a, at::Tensor b_nt, at::Tensor c, double alpha) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && a.stride(1) == 1);
  TORCH_CHECK(b_nt.dim() == 2 && b_nt.stride(1) == 1,
              "B must be K-contiguous [N, K] (pass weight / b.t())");
  TORCH_CHECK(c.dim() == 2 && c.stride(1) == 1);
  int M = a.size(0), K = a.size(1), N = b_nt.size(0);
  TORCH_CHECK(b_nt.size(1) == K && c.size(0) == M && c.size(1) == N);
  // v2 (256^2 tile, global_load_lds pipeline) needs K % 64 == 0 and 16 B
  // aligned rows; otherwise the general 128^2 kernel handles it.
  bool v2_ok = dtype_code(a) == 0 && (K % 64 == 0) && (a.stride(0) % 8 == 0) &&
               (b_nt.stride(0) % 8 == 0) && M > 128;
  if (v2_ok) {
    check_hip(fi_gemm_nt_v2(a.data_ptr(), b_nt.data_ptr(), c.data_ptr(), M, N, K,
                            a.stride(0), b_nt.stride(0), c.stride(0), (float)alpha,
                            cur_stream(a)),
              "fi_gemm_nt_v2");
    return;
  }
  check_hip(fi_gemm_nt(dtype_code(a), a.data_ptr(), b_nt.data_ptr(), c.data_ptr(), M, N,
                       K, a.stride(0), b_nt.stride(0), c.stride(0), (float)alpha,
                       cur_stream(a)),
            "fi_gemm_nt");
}

// grouped NT GEMM: a [M,K]; w [num_w, N, K]; c [M,N]; m_indptr [S+1] device.
void group_gemm_nt(at::Tensor a, at::Tensor w, at::Tensor c, at::Tensor m_indptr,
                   c10::optional<at::Tensor> w_indices, int64_t max_m_tiles) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && a.stride(1) == 1);
  TORCH_CHECK(w.dim() == 3 && w.stride(2) == 1, "weights must be [E, N, K] K-contig");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16);
  int K = a.size(1), N = w.size(1);
  TORCH_CHECK(w.size(2) == K && c.size(1) == N);
  int S = m_indptr.size(0) - 1;
  check_hip(fi_group_gemm_nt(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), m_indptr.data_ptr<int32_t>(),
                w_indices.has_value() ? w_indices->data_ptr<int32_t>() : nullptr, S,
                (int)max_m_tiles, N, K, a.stride(0), w.stride(1), w.stride(0),
                c.stride(0), cur_stream(a)),
            "fi_group_gemm_nt");
}

// fp8 grouped/groupwise GEMM. a/w are uint8 (fp8 e4m3) tensors.
void gemm_fp8_grouped(at::Tensor a, at::Tensor w, at::Tensor c, at::Tensor m_indptr,
                      c10::optional<at::Tensor> w_indices, int64_t max_m_tiles,
                      c10::optional<at::Tensor> a_scales,
                      c10::optional<at::Tensor> b_scales, double scalar_scale,
                      int64_t flat_tiles = 0) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && a.stride(1) == 1);
  TORCH_CHECK(w.dim() == 3 && w.stride(2) == 1);
  TORCH_CHECK(c.scalar_type() == at::kBFloat16);
  int K = a.size(1), N = w.size(1);
  int S = m_indptr.size(0) - 1;
  // MX mode: u8 e8m0 scale tensors (per-row [M, K/128] + per-block
  // [S, K/128, N/128] bytes) select the hardware-scaled f8f6f4 path
  bool mx = a_scales.has_value() && a_scales->scalar_type() == at::kByte;
  int64_t a_scale_stride = a_scales.has_value() ? a_scales->stride(0) : 0;
  check_hip(fi_gemm_fp8_grouped(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), m_indptr.data_ptr<int32_t>(),
                w_indices.has_value() ? w_indices->data_ptr<int32_t>() : nullptr, S,
                (int)max_m_tiles, N, K, a.stride(0), w.stride(1), w.stride(0),
                c.stride(0),
                a_scales.has_value() ? (float*)a_scales->data_ptr() : nullptr,
                b_scales.has_value() ? (float*)b_scales->data_ptr() : nullptr,
                (float)scalar_scale, a_scale_stride, (int)flat_tiles, mx ? 1 : 0,
                cur_stream(a)),
            "fi_gemm_fp8_grouped");
}

// per-128-group fp8 quantization: returns (q uint8 [rows,K], scale f32)
void per_group_quant_fp8(at::Tensor x, at::Tensor q, at::Tensor scale,
                         bool trans_scale, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.stride(1) == 1);
  check_hip(fi_per_group_quant_fp8(dtype_code(x), trans_scale ? 1 : 0, x.data_ptr(),
                                   q.data_ptr(), scale.data_ptr<float>(), x.size(0),
                                   x.size(1), x.stride(0), (float)eps, cur_stream(x)),
            "fi_per_group_quant_fp8");
}

void scale_quant_fp8(at::Tensor x, at::Tensor q, at::Tensor inv_scale) {
  check_hip(fi_scale_quant_fp8(dtype_code(x), x.data_ptr(), q.data_ptr(),
                               inv_scale.data_ptr<float>(), x.numel(), cur_stream(x)),
            "fi_scale_quant_fp8");
}

void topk_op(at::Tensor x, c10::optional<at::Tensor> out_v, at::Tensor out_i,
             int64_t k, c10::optional<at::Tensor> lengths,
             c10::optional<at::Tensor> row_starts, c10::optional<at::Tensor> offsets,
             c10::optional<at::Tensor> page_table,
             c10::optional<at::Tensor> row_to_batch,
             c10::optional<at::Tensor> page_table_row_starts, int64_t page_size,
             int64_t mode, int64_t tie_break) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.scalar_type() == at::kFloat);
#define IPTR(t) (t.has_value() ? t->data_ptr<int32_t>() : nullptr)
  check_hip(fi_topk(x.data_ptr<float>(),
                    out_v.has_value() ? out_v->data_ptr<float>() : nullptr,
                    out_i.data_ptr<int32_t>(), IPTR(lengths), IPTR(row_starts),
                    IPTR(offsets), IPTR(page_table), IPTR(row_to_batch),
                    IPTR(page_table_row_starts),
                    page_table.has_value() ? (int)page_table->size(1) : 0,
                    (int)page_size, (int)mode, x.size(0), x.size(1), (int)k,
                    x.stride(0), (int)tie_break, cur_stream(x)),
            "fi_topk");
#undef IPTR
}

void packbits_op(at::Tensor x, at::Tensor y) {
  check_hip(fi_packbits(x.data_ptr<uint8_t>(), y.data_ptr<uint8_t>(), x.numel(),
                        cur_stream(x)),
            "fi_packbits");
}

void segment_packbits_op(at::Tensor x, at::Tensor y, at::Tensor x_indptr,
                         at::Tensor y_indptr) {
  check_hip(fi_segment_packbits(x.data_ptr<uint8_t>(), y.data_ptr<uint8_t>(),
                                x_indptr.data_ptr<int32_t>(),
                                y_indptr.data_ptr<int32_t>(), x_indptr.size(0) - 1,
                                cur_stream(x)),
            "fi_segment_packbits");
}

void selective_state_update(at::Tensor state, at::Tensor x, at::Tensor dt,
                            at::Tensor A, at::Tensor B, at::Tensor C,
                            c10::optional<at::Tensor> D, c10::optional<at::Tensor> z,
                            c10::optional<at::Tensor> dt_bias, at::Tensor out,
                            bool dt_softplus) {
  TORCH_CHECK(state.is_cuda() && state.dim() == 4 && state.is_contiguous());
  fi_ext::SSUParams p{};
  p.state = state.data_ptr();
  p.x = x.data_ptr();
  p.dt = dt.data_ptr();
  p.A = A.data_ptr();
  p.Bm = B.data_ptr();
  p.Cm = C.data_ptr();
  p.D = D.has_value() ? D->data_ptr() : nullptr;
  p.z = z.has_value() ? z->data_ptr() : nullptr;
  p.dt_bias = dt_bias.has_value() ? dt_bias->data_ptr() : nullptr;
  p.out = out.data_ptr();
  p.batch = state.size(0);
  p.nheads = state.size(1);
  p.headdim = state.size(2);
  p.dstate = state.size(3);
  p.ngroups = B.size(1);
  p.dt_softplus = dt_softplus ? 1 : 0;
  p.state_f32 = state.scalar_type() == at::kFloat && x.scalar_type() != at::kFloat;
  check_hip(fi_selective_state_update(dtype_code(x), &p, cur_stream(x)),
            "fi_selective_state_update");
}

void gather_rows(at::Tensor src, at::Tensor dst, at::Tensor row_map) {
  check_hip(fi_gather_rows(dtype_code(src), src.data_ptr(), dst.data_ptr(),
                           row_map.data_ptr<int32_t>(), dst.size(0), src.size(1),
                           cur_stream(src)),
            "fi_gather_rows");
}

void moe_finalize(at::Tensor h, at::Tensor out, at::Tensor pos, at::Tensor w) {
  check_hip(fi_moe_finalize(dtype_code(h), h.data_ptr(), out.data_ptr(),
                            pos.data_ptr<int32_t>(), w.data_ptr<float>(), out.size(0),
                            pos.size(1), h.size(1), cur_stream(h)),
            "fi_moe_finalize");
}

void gdn_decode(at::Tensor state, at::Tensor q, at::Tensor k, at::Tensor v,
                at::Tensor g, at::Tensor beta, at::Tensor out) {
  TORCH_CHECK(state.is_cuda() && state.dim() == 4 && state.is_contiguous());
  int B = state.size(0), H = state.size(1), Dk = state.size(2), Dv = state.size(3);
  bool state_f32 =
      state.scalar_type() == at::kFloat && q.scalar_type() != at::kFloat;
  bool per_channel = g.dim() == 3;
  check_hip(fi_gdn_decode(dtype_code(q), state_f32, per_channel, state.data_ptr(),
                          q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          g.data_ptr<float>(), beta.data_ptr<float>(), out.data_ptr(),
                          B, H, Dk, Dv, cur_stream(q)),
            "fi_gdn_decode");
}

void ipc_memcpy_to(int64_t dst_ptr, at::Tensor src) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous());
  check_hip(hipMemcpyAsync((void*)dst_ptr, src.data_ptr(),
                           src.numel() * src.element_size(),
                           hipMemcpyDeviceToDevice, cur_stream(src)),
            "hipMemcpyAsync");
}

static fi::ARParams make_ar_params(const std::vector<int64_t>& bufs, int64_t rank,
                                   int64_t seq, at::Tensor error_flag,
                                   int64_t spin_limit, int64_t data_off) {
  fi::ARParams p{};
  TORCH_CHECK((int)bufs.size() <= fi::kMaxRanks);
  for (size_t i = 0; i < bufs.size(); ++i) p.bufs[i] = (unsigned long long)bufs[i];
  p.data_off = (unsigned long long)data_off;
  p.world = (int)bufs.size();
  p.rank = (int)rank;
  p.seq = (unsigned long long)seq;
  p.spin_limit = (unsigned long long)spin_limit;
  p.error_flag = error_flag.data_ptr<int32_t>();
  return p;
}

void one_shot_all_reduce(at::Tensor out, std::vector<int64_t> bufs, int64_t rank,
                         int64_t seq, at::Tensor error_flag, int64_t spin_limit,
                         int64_t data_off) {
  fi::ARParams p = make_ar_params(bufs, rank, seq, error_flag, spin_limit, data_off);
  check_hip(fi_one_shot_ar(dtype_code(out), &p, out.data_ptr(), out.numel(),
                           cur_stream(out)),
            "fi_one_shot_ar");
}

void two_shot_all_reduce(at::Tensor out, std::vector<int64_t> bufs, int64_t rank,
                         int64_t seq, at::Tensor error_flag, int64_t spin_limit,
                         int64_t data_off, int64_t shard_off) {
  fi::ARParams p = make_ar_params(bufs, rank, seq, error_flag, spin_limit, data_off);
  check_hip(fi_two_shot_ar(dtype_code(out), &p, (unsigned long long)shard_off,
                           out.data_ptr(), out.numel(), cur_stream(out)),
            "fi_two_shot_ar");
}

void one_shot_all_reduce_rmsnorm(at::Tensor out, c10::optional<at::Tensor> residual,
                                 at::Tensor weight, std::vector<int64_t> bufs,
                                 int64_t rank, int64_t seq, at::Tensor error_flag,
                                 int64_t spin_limit, int64_t data_off, double eps) {
  fi::ARParams p = make_ar_params(bufs, rank, seq, error_flag, spin_limit, data_off);
  int d = out.size(-1);
  int rows = out.numel() / d;
  check_hip(fi_one_shot_ar_rmsnorm(dtype_code(out), &p, out.data_ptr(),
                                   residual ? residual->data_ptr() : nullptr,
                                   weight.data_ptr(), rows, d, (float)eps,
                                   cur_stream(out)),
            "fi_one_shot_ar_rmsnorm");
}

void qk_rope(at::Tensor qkv, at::Tensor qw, at::Tensor kw, at::Tensor cos_t,
             at::Tensor sin_t, at::Tensor q_out, at::Tensor k_out,
             at::Tensor v_out, int64_t S, int64_t Hq, int64_t Hk, int64_t Hv,
             int64_t D, double eps, double attn_factor, bool qk_norm) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous());
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && sin_t.scalar_type() == at::kFloat);
  int64_t tokens = qkv.numel() / ((Hq + Hk + Hv) * D);
  check_hip(fi_qk_rope(dtype_code(qkv), qkv.data_ptr(), qw.data_ptr(),
                       kw.data_ptr(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), q_out.data_ptr(), k_out.data_ptr(),
                       v_out.data_ptr(), tokens, (int)S, (int)Hq, (int)Hk, (int)Hv,
                       (int)D, (float)eps, (float)attn_factor, qk_norm,
                       cur_stream(qkv)),
            "fi_qk_rope");
}

void mhc_post(at::Tensor x, at::Tensor residual, at::Tensor post_mix,
              at::Tensor comb_mix, at::Tensor out) {
  TORCH_CHECK(x.is_cuda() && residual.dim() == 3 && residual.size(1) == 4);
  int64_t tokens = residual.size(0);
  int H = residual.size(2);
  check_hip(fi_mhc_post(dtype_code(x), x.data_ptr(), residual.data_ptr(),
                        post_mix.data_ptr<float>(), comb_mix.data_ptr<float>(),
                        out.data_ptr(), tokens, H, cur_stream(x)),
            "fi_mhc_post");
}

void mhc_pre(at::Tensor dot_mix, c10::optional<at::Tensor> sqrsum,
             at::Tensor residual, at::Tensor scale, at::Tensor base,
             at::Tensor post_mix, at::Tensor comb_mix, at::Tensor layer_input,
             int64_t num_splits, double inv_k, double rms_eps, double pre_eps,
             double sink_eps, double post_mult, int64_t sink_repeat) {
  TORCH_CHECK(residual.is_cuda() && residual.dim() == 3 && residual.size(1) == 4);
  int64_t tokens = residual.size(0);
  int H = residual.size(2);
  check_hip(fi_mhc_pre(dtype_code(residual), !sqrsum.has_value(),
                       dot_mix.data_ptr<float>(),
                       sqrsum ? sqrsum->data_ptr<float>() : nullptr,
                       residual.data_ptr(), scale.data_ptr<float>(),
                       base.data_ptr<float>(), post_mix.data_ptr<float>(),
                       comb_mix.data_ptr<float>(), layer_input.data_ptr(), tokens, H,
                       (int)num_splits, (float)inv_k, (float)rms_eps, (float)pre_eps,
                       (float)sink_eps, (float)post_mult, (int)sink_repeat,
                       cur_stream(residual)),
            "fi_mhc_pre");
}

void concat_mla_k(at::Tensor k, at::Tensor k_nope, at::Tensor k_rope) {
  TORCH_CHECK(k.is_cuda() && k.dim() == 3 && k_nope.dim() == 3 && k_rope.dim() == 3);
  int64_t tokens = k.size(0);
  int Hk = k.size(1), nope = k_nope.size(2), rope = k_rope.size(2);
  TORCH_CHECK(k.size(2) == nope + rope && k_rope.size(1) == 1);
  check_hip(fi_concat_mla_k(dtype_code(k), k.data_ptr(), k_nope.data_ptr(),
                            k_rope.data_ptr(), tokens, Hk, nope, rope,
                            cur_stream(k)),
            "fi_concat_mla_k");
}

void ssd_scan(at::Tensor x, at::Tensor dt, at::Tensor A, at::Tensor B, at::Tensor C,
              std::optional<at::Tensor> D, std::optional<at::Tensor> z,
              std::optional<at::Tensor> dt_bias,
              std::optional<at::Tensor> init_states,
              std::optional<at::Tensor> final_states, at::Tensor out,
              bool dt_softplus, double dt_min, double dt_max) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  TORCH_CHECK(A.scalar_type() == at::kFloat && dt.scalar_type() == at::kFloat);
  int batch = x.size(0), L = x.size(1), H = x.size(2), P = x.size(3);
  int G = B.size(2), dstate = B.size(3);
  bool d_has_hdim = D && D->dim() == 2;
  check_hip(
      fi_ssd_scan(dtype_code(x), x.data_ptr(), dt.data_ptr<float>(),
                  A.data_ptr<float>(), B.data_ptr(), C.data_ptr(),
                  D ? D->data_ptr<float>() : nullptr, z ? z->data_ptr() : nullptr,
                  dt_bias ? dt_bias->data_ptr<float>() : nullptr,
                  init_states ? init_states->data_ptr<float>() : nullptr,
                  final_states ? final_states->data_ptr<float>() : nullptr,
                  out.data_ptr(), batch, L, H, G, P, dstate, dt_softplus,
                  (float)dt_min, (float)dt_max, d_has_hdim, cur_stream(x)),
      "fi_ssd_scan");
}

void gdn_chunk(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor gate,
               at::Tensor beta, at::Tensor out, at::Tensor cu_seqlens,
               std::optional<at::Tensor> init_state,
               std::optional<at::Tensor> final_state, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3 && q.is_contiguous());
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt);
  int H = q.size(1), D = q.size(2);
  int num_seqs = cu_seqlens.size(0) - 1;
  bool per_channel = gate.dim() == 3;
  check_hip(fi_gdn_chunk(dtype_code(q), per_channel, q.data_ptr(), k.data_ptr(),
                         v.data_ptr(), gate.data_ptr<float>(), beta.data_ptr<float>(),
                         out.data_ptr(), cu_seqlens.data_ptr<int32_t>(),
                         init_state ? init_state->data_ptr<float>() : nullptr,
                         final_state ? final_state->data_ptr<float>() : nullptr,
                         (float)scale, num_seqs, H, D, cur_stream(q)),
            "fi_gdn_chunk");
}

// fastdiv self-check (host): returns n // d computed via the magic scheme.
std::vector<int64_t> debug_fastdiv(int64_t d, std::vector<int64_t> ns) {
  fi::uint_fastdiv fd((uint32_t)d);
  std::vector<int64_t> out;
  for (auto n64 : ns) out.push_back(fd.div((uint32_t)n64));
  return out;
}

}  // namespace


void moe_topk_softmax_run(at::Tensor logits, at::Tensor weights, at::Tensor ids,
                          bool renorm) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kFloat);
  check_hip(fi_moe_topk_softmax(logits.data_ptr<float>(), weights.data_ptr<float>(),
                                ids.data_ptr<int32_t>(), logits.size(0),
                                logits.size(1), ids.size(1), renorm ? 1 : 0,
                                cur_stream(logits)),
            "fi_moe_topk_softmax");
}

void dsv3_routing_run(at::Tensor logits, c10::optional<at::Tensor> bias,
                      at::Tensor weights, at::Tensor ids, int64_t n_group,
                      int64_t topk_group, double scale) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kFloat);
  check_hip(fi_dsv3_routing(logits.data_ptr<float>(),
                            bias.has_value() ? bias->data_ptr<float>() : nullptr,
                            weights.data_ptr<float>(), ids.data_ptr<int32_t>(),
                            logits.size(0), logits.size(1), ids.size(1),
                            (int)n_group, (int)topk_group, (float)scale,
                            cur_stream(logits)),
            "fi_dsv3_routing");
}

void moe_build_permute_run(at::Tensor ids, at::Tensor counts, at::Tensor m_indptr,
                           at::Tensor cursor, at::Tensor token_of_copy,
                           at::Tensor inv, int64_t align) {
  int n = ids.numel(), k = ids.size(-1), E = counts.size(0);
  check_hip(fi_moe_build_permute(ids.data_ptr<int32_t>(), counts.data_ptr<int32_t>(),
                                 m_indptr.data_ptr<int32_t>(),
                                 cursor.data_ptr<int32_t>(),
                                 token_of_copy.data_ptr<int32_t>(),
                                 inv.data_ptr<int32_t>(), n, k, E, (int)align,
                                 cur_stream(ids)),
            "fi_moe_build_permute");
}

void gather_quant_run(at::Tensor src, at::Tensor token_of_copy, at::Tensor dst,
                      at::Tensor scale) {
  // e8m0 mode is selected by the scale tensor's dtype (u8 = MX bytes)
  bool e8m0 = scale.scalar_type() == at::kByte;
  check_hip(fi_gather_quant(dtype_code(src), src.data_ptr(),
                            token_of_copy.data_ptr<int32_t>(),
                            dst.data_ptr<uint8_t>(), (float*)scale.data_ptr(),
                            dst.size(0), dst.size(1), e8m0 ? 1 : 0,
                            cur_stream(src)),
            "fi_gather_quant");
}

void silu_mul_quant_run(at::Tensor h, at::Tensor dst, at::Tensor scale, bool gelu) {
  bool e8m0 = scale.scalar_type() == at::kByte;
  check_hip(fi_silu_mul_quant(dtype_code(h), h.data_ptr(), dst.data_ptr<uint8_t>(),
                              (float*)scale.data_ptr(), dst.size(0), dst.size(1),
                              gelu ? 1 : 0, e8m0 ? 1 : 0, cur_stream(h)),
            "fi_silu_mul_quant");
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm);
  m.def("rmsnorm_silu", &rmsnorm_silu);
  m.def("rmsnorm_quant", &rmsnorm_quant);
  m.def("layernorm_quant", &layernorm_quant);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("layernorm", &layernorm);
  m.def("act_and_mul", &act_and_mul);
  m.def("apply_rope", &apply_rope);
  m.def("append_paged_kv_cache", &append_paged_kv_cache);
  m.def("get_batch_indices_positions", &get_batch_indices_positions);
  m.def("merge_states", &merge_states);
  m.def("moe_topk_softmax_run", &moe_topk_softmax_run);
  m.def("dsv3_routing_run", &dsv3_routing_run);
  m.def("moe_build_permute_run", &moe_build_permute_run);
  m.def("gather_quant_run", &gather_quant_run);
  m.def("silu_mul_quant_run", &silu_mul_quant_run);
  m.def("merge_state_in_place", &merge_state_in_place);
  m.def("batch_decode_run", &batch_decode_run);
  m.def("batch_decode_fused_run", &batch_decode_fused_run);
  m.def("batch_decode_mfma_run", &batch_decode_mfma_run);
  m.def("batch_attention_run", &batch_attention_run);
  m.def("gemm_nt", &gemm_nt);
  m.def("batch_prefill_run", &batch_prefill_run);
  m.def("softmax", &softmax_op);
  m.def("sampling", &sampling_op);
  m.def("renorm", &renorm_op);
  m.def("chain_speculative", &chain_speculative_op);
  m.def("mla_run", &mla_run);
  m.def("group_gemm_nt", &group_gemm_nt);
  m.def("gemm_fp8_grouped", &gemm_fp8_grouped);
  m.def("per_group_quant_fp8", &per_group_quant_fp8);
  m.def("scale_quant_fp8", &scale_quant_fp8);
  m.def("topk", &topk_op);
  m.def("packbits", &packbits_op);
  m.def("segment_packbits", &segment_packbits_op);
  m.def("selective_state_update", &selective_state_update);
  m.def("gather_rows", &gather_rows);
  m.def("moe_finalize", &moe_finalize);
  m.def("gdn_decode", &gdn_decode);
  m.def("gdn_chunk", &gdn_chunk);
  m.def("ssd_scan", &ssd_scan);
  m.def("mhc_post", &mhc_post);
  m.def("qk_rope", &qk_rope);
  m.def("mhc_pre", &mhc_pre);
  m.def("concat_mla_k", &concat_mla_k);
  m.def("ipc_memcpy_to", &ipc_memcpy_to);
  m.def("one_shot_all_reduce", &one_shot_all_reduce);
  m.def("two_shot_all_reduce", &two_shot_all_reduce);
  m.def("one_shot_all_reduce_rmsnorm", &one_shot_all_reduce_rmsnorm);
  m.def("debug_fastdiv", &debug_fastdiv);
}
