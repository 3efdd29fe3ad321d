// Batch prefill attention (FA2-class) for gfx950 — the flagship kernel.
// Functional parity with reference include/flashinfer/attention/prefill.cuh
// (BatchPrefillWithPagedKVCacheKernel:4134, Ragged:2746, single:2511) but a
// from-scratch CDNA4 design (guide §B "fused attention prefill" structure):
//
//  * 4 waves x 32 packed q-rows = CTA_TILE_Q 128 (GQA rows packed q_pos x
//    group, reference scheduler packing), KV tile 64.
//  * swapped QK^T: S^T = mfma32(K_frag, Q_frag) so each lane owns ONE q-row's
//    logits (col = lane&31) — the online softmax is 16 in-lane values + one
//    shfl_xor(32) exchange, no cross-lane reduction tree.
//  * O accumulated TRANSPOSED (O^T = mfma32(V^T_frag, P^T_frag)): the
//    per-q-row rescale factor is a per-lane scalar.
//  * K staged row-major [64][D] with XOR swizzle (32-way bank conflict
//    without it, guide §6 G4); V staged transposed [D][64] with the same
//    swizzle so PV A-fragments are contiguous 16 B ds reads.
//  * base-2 softmax domain (v_exp_f32 is exp2 natively): scale2 folds
//    log2(e); lse output is directly base-2 (library convention).
//  * causal / sliding-window / logits-soft-cap masking in-register.
//  * paged (page table) or ragged (contiguous) KV via template.
// Work items (req, q_tile) are host-planned; grid (n_tiles, num_kv_heads).
#include "fi/prefill_body.hpp"

namespace fi {

template <typename T, typename TKV, int HDQK, int HDVO, int CTAQ, bool PAGED,
          bool CAUSAL, bool MASK = false>
__global__ __launch_bounds__(CTAQ * 2, ((HDQK >= 192 || HDVO >= 256) ? 1 : 2)) void batch_prefill_kernel(PrefillParams p) {
  __shared__ char smem[prefill_tile_smem_bytes<T, HDQK, HDVO>()];
  const int tile = blockIdx.x;
  if (tile >= p.n_tiles) return;
  prefill_tile_body<T, TKV, HDQK, HDVO, CTAQ, PAGED, CAUSAL, MASK>(
      p, p.tile_req[tile], p.tile_qstart[tile], blockIdx.y,
      p.tile_kv_chunk ? (int)p.tile_kv_chunk[tile] : -1, smem);
  // chained second short tile (same WG, fresh state; the barrier orders the
  // first item's epilogue LDS reads against the second's Q staging)
  if (p.tile_req_b && p.tile_req_b[tile] >= 0) {
    __syncthreads();
    prefill_tile_body<T, TKV, HDQK, HDVO, CTAQ, PAGED, CAUSAL, MASK>(
        p, p.tile_req_b[tile], p.tile_qstart_b[tile], blockIdx.y, -1, smem);
  }
}

template <typename T, typename TKV>
hipError_t prefill_dispatch(PrefillParams& p, bool paged, hipStream_t stream) {
  dim3 g(p.n_tiles, p.num_kv_heads), blk(p.cta_q * 2);
#define LAUNCH_PF2(HD, HV, CQ, PG, CS)                                            \
  hipLaunchKernelGGL((batch_prefill_kernel<T, TKV, HD, HV, CQ, PG, CS>), g, blk, \
                     0, stream, p)
#define LAUNCH_PF(HD, HV, PG, CS)                        \
  do {                                                   \
    if (p.cta_q == 256) LAUNCH_PF2(HD, HV, 256, PG, CS); \
    else LAUNCH_PF2(HD, HV, 128, PG, CS);                \
  } while (0)
#define LAUNCH_PFM(HD, HV, CQ, PG)                                                \
  hipLaunchKernelGGL((batch_prefill_kernel<T, TKV, HD, HV, CQ, PG, false, true>), \
                     g, blk, 0, stream, p)
#define DISPATCH_PC(HD, HV)                                     \
  do {                                                          \
    if (p.mask_data) {                                          \
      if (paged) {                                              \
        if (p.cta_q == 256) LAUNCH_PFM(HD, HV, 256, true);      \
        else LAUNCH_PFM(HD, HV, 128, true);                     \
      } else {                                                  \
        if (p.cta_q == 256) LAUNCH_PFM(HD, HV, 256, false);     \
        else LAUNCH_PFM(HD, HV, 128, false);                    \
      }                                                         \
    } else if (paged) {                                         \
      if (p.causal) LAUNCH_PF(HD, HV, true, true);              \
      else LAUNCH_PF(HD, HV, true, false);                      \
    } else {                                                    \
      if (p.causal) LAUNCH_PF(HD, HV, false, true);             \
      else LAUNCH_PF(HD, HV, false, false);                     \
    }                                                           \
  } while (0)
  int hv = p.head_dim_vo > 0 ? p.head_dim_vo : p.head_dim;
  if (p.head_dim == hv) {
    switch (p.head_dim) {
      case 64: DISPATCH_PC(64, 64); break;
      case 128: DISPATCH_PC(128, 128); break;
      case 256: DISPATCH_PC(256, 256); break;
      default: return hipErrorInvalidValue;
    }
  } else if (p.head_dim == 192 && hv == 128) {
    // DeepSeek MHA prefill (q/k 128 nope + 64 rope, v 128)
    DISPATCH_PC(192, 128);
  } else {
    return hipErrorInvalidValue;
  }
#undef DISPATCH_PC
#undef LAUNCH_PF
#undef LAUNCH_PF2
#undef LAUNCH_PFM
  return hipGetLastError();
}

}  // namespace fi

extern "C" hipError_t fi_batch_prefill(int dtype, int kv_dtype, fi::PrefillParams* p,
                                       int paged, hipStream_t stream) {
  if (p->n_tiles == 0) return hipSuccess;
  if (kv_dtype == dtype) {
    switch (dtype) {
      case 0: return fi::prefill_dispatch<fi::bf16, fi::bf16>(*p, paged, stream);
      case 1: return fi::prefill_dispatch<fi::fp16, fi::fp16>(*p, paged, stream);
    }
  } else if (kv_dtype == 3 && dtype == 0) {
    return fi::prefill_dispatch<fi::bf16, fi::fp8_e4m3>(*p, paged, stream);
  }
  return hipErrorInvalidValue;
}
