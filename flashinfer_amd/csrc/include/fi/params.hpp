// Kernel parameter structs shared between the torch binding TU and the HIP
// kernel TUs (single definition — no mirrored layouts).
#pragma once
#include <stdint.h>

#include "fi/fastdiv.hpp"

namespace fi {

struct RopeParams {
  const void* q;  // [nnz, Hq, D]
  const void* k;  // [nnz, Hkv, D]
  void* q_out;
  void* k_out;
  const int32_t* pos_ids;      // [nnz]
  const float* cos_sin_cache;  // [max_pos, rot_dim] or null
  int64_t nnz;
  int num_qo_heads, num_kv_heads;
  int head_dim, rot_dim;
  int64_t q_stride_n, q_stride_h, k_stride_n, k_stride_h;
  int64_t o_q_stride_n, o_q_stride_h, o_k_stride_n, o_k_stride_h;
  float rope_rcp_scale;
  float rope_theta;
  float smooth_a, smooth_b, rcp_factor;
  bool interleave;
};

struct DecodeParams {
  const void* q;  // [n_req, Hq, D]
  void* k_data;
  void* v_data;
  const int32_t* kv_indices;
  const int32_t* kv_indptr;
  const int32_t* kv_last_page_len;
  uint_fastdiv page_size;
  int num_kv_heads, num_qo_heads, head_dim;
  int64_t stride_page, stride_n, stride_h;
  const int32_t* work_req;    // [n_items]
  const int32_t* work_chunk;  // [n_items]
  int n_items;
  int chunk_size;
  float* tmp_v;  // [n_items, Hq, D]
  float* tmp_s;  // [n_items, Hq]
  int64_t q_stride_n, q_stride_h;
  float sm_scale;
  float logits_soft_cap;  // 0 = disabled
  int window_left;        // -1 = disabled
  int alibi;              // ALiBi position bias
  // fused whole-request path (no split/merge): direct output in q dtype
  void* o;                // [batch, Hq, D]
  float* lse;             // [batch, Hq] base-2, optional
  int64_t o_stride_n, o_stride_h;
  int batch;
  // MFMA decode cross-WG split (gridDim.z = split): partials + monotonic
  // arrival counters (modulo split — no per-launch zeroing, hipGraph-safe)
  int split;
  uint32_t* counters;     // [batch * num_kv_heads]
};

struct PrefillParams {
  const void* q;  // [nnz_q, Hq, D]
  void* out;      // [nnz_q, Hq, D]
  float* lse;     // [nnz_q, Hq] base-2, optional
  const int32_t* qo_indptr;  // [batch+1] token offsets
  void* k_data;
  void* v_data;
  const int32_t* kv_indices;
  const int32_t* kv_indptr;  // paged: page ranges; ragged: token offsets
  const int32_t* kv_last_page_len;
  uint_fastdiv page_size;
  int64_t kv_stride_page, kv_stride_n, kv_stride_h;
  const int32_t* tile_req;     // [n_tiles]
  const int32_t* tile_qstart;  // [n_tiles] packed-row offset in request
  // optional second item per WG (causal short-tile chaining; -1 = none):
  // halves the dispatch-round count when most tiles are short-diagonal
  const int32_t* tile_req_b;
  const int32_t* tile_qstart_b;
  int n_tiles;
  int num_qo_heads, num_kv_heads, head_dim;
  uint_fastdiv group;  // Hq / Hkv
  int64_t q_stride_n, q_stride_h, o_stride_n, o_stride_h;
  float sm_scale;
  float logits_soft_cap;  // 0 disabled
  int window_left;        // -1 disabled
  int causal;
  int cta_q;              // 128 or 256 packed q rows per tile
  const uint8_t* mask_data;        // packed bitmask (little), or null
  const int32_t* mask_byte_indptr; // per-request byte offset into mask_data
  int alibi;                       // ALiBi position bias (slope by qo head)
  float k_descale, v_descale;      // fp8-KV dequant factors (staging)
  unsigned long long* prof_buf;    // intra-kernel event buffer (optional)
  // head_dim_qk != head_dim_vo (DeepSeek MHA 192/128): V gets its own strides
  int head_dim_vo;                 // = head_dim when square
  int64_t v_stride_page, v_stride_n, v_stride_h;
  // split-KV (reference scheduler.cuh:545 PrefillSplitQOKVIndptr role):
  // when tile_kv_chunk != null each tile covers one kv chunk and writes
  // normalized f32 partials + base-2 lse to (tmp_v, tmp_s) at
  // slot = req_slot_base[req] + qpos * n_chunks(req) + chunk; the host then
  // runs the LSE merge kernel over per-token slot ranges.
  const int32_t* tile_kv_chunk;    // [n_tiles] chunk index, null = no split
  int kv_chunk;                    // tokens per kv chunk
  const int32_t* req_slot_base;    // [batch]
  float* tmp_v;                    // [n_slots, Hq, head_dim_vo]
  float* tmp_s;                    // [n_slots, Hq]
};


// one-shot hipIpc allreduce (csrc/comm_ar.hip)
constexpr int kMaxRanks = 8;
struct ARParams {
  unsigned long long bufs[kMaxRanks];  // device pointers (rank-local mapping)
  unsigned long long data_off;  // double-buffer slot offset (seq parity)
  int world, rank;
  unsigned long long seq;
  unsigned long long spin_limit;
  int* error_flag;  // device int: set to 1 on spin timeout
};

struct SamplingParams {
  const float* probs;     // [rows, vocab] (or logits)
  float* out_probs;       // optional output distribution
  int32_t* out_ids;       // [rows] sampled token
  const float* uniforms;  // [rows, rounds]
  const float* top_p;     // [rows] or null (then scalar_p)
  const int32_t* top_k;   // [rows] or null (then scalar_k)
  const int32_t* row_indices;  // optional indirection probs row = row_indices[b]
  float scalar_p;
  int scalar_k;
  float min_p;
  int rows, vocab, rounds;
  uint64_t stride_row;
  float temperature;  // for softmax
};


struct SpecParams {
  const float* draft_probs;   // [B, n, V]
  const int32_t* draft_ids;   // [B, n]
  const float* target_probs;  // [B, n+1, V]
  int32_t* out_ids;           // [B, n+1]
  int32_t* accepted_num;      // [B] += accepted count
  int32_t* emitted_num;       // [B] += emitted draft tokens
  const float* uniforms;      // [B, n+1+rounds]
  int B, n, vocab;
};


struct MlaParams {
  const void* q_nope;  // [nnz, H, 512]
  const void* q_pe;    // [nnz, H, 64]
  void* ckv_data;      // [pages, page_size, 512]
  void* kpe_data;      // [pages, page_size, 64]
  const int32_t* qo_indptr;
  const int32_t* kv_indices;
  const int32_t* kv_indptr;
  const int32_t* kv_last_page_len;
  uint_fastdiv page_size;
  uint_fastdiv num_heads_fd;
  int64_t ckv_stride_page, ckv_stride_n;
  int64_t kpe_stride_page, kpe_stride_n;
  int64_t q_nope_stride_n, q_nope_stride_h, q_pe_stride_n, q_pe_stride_h;
  const int32_t* tile_req;    // [n_items]
  const int32_t* tile_row0;   // [n_items] packed-row offset
  const int32_t* tile_chunk;  // [n_items]
  int n_items;
  int chunk_size;
  int max_chunks;
  int num_heads;
  float* tmp_v;  // [nnz*H, max_chunks, 512]
  float* tmp_s;  // [nnz*H, max_chunks]
  float sm_scale;
  int causal;
  float ckv_descale, kpe_descale;  // fp8-KV dequant factors (staging)
};

struct SSUParams {
  void* state;        // [B, H, P, S] (f32 or T)
  const void* x;      // [B, H, P]
  const void* dt;     // [B, H]
  const void* A;      // [H]
  const void* Bm;     // [B, G, S]
  const void* Cm;     // [B, G, S]
  const void* D;      // [H] or null
  const void* z;      // [B, H, P] or null
  const void* dt_bias;  // [H] or null
  void* out;          // [B, H, P]
  int batch, nheads, headdim, dstate, ngroups;
  int dt_softplus;
  int state_f32;  // state stored as f32 (else T)
};


}  // namespace fi
