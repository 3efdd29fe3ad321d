// Persistent holistic BatchAttention — mixed prefill+decode batches in ONE
// kernel launch (reference include/flashinfer/attention/persistent.cuh +
// persistent_template.cuh:39 + plan scheduler.cuh:1243 TwoStageHolisticPlan
// role, re-designed for CDNA4):
//
//  * grid = one workgroup per CU (persistent), 512 threads (8 waves);
//  * an ATOMIC work-queue ticket hands out host-planned items, each tagged
//    PREFILL (req, qstart, kv_head: a 256-row FA2 MFMA tile — the flagship
//    prefill pipeline) or DECODE (req, kv_head: a 32-row MFMA decode item
//    with the GQA group as the q dimension — 8 barrier-free waves split the
//    KV and merge through LDS). The two tile sizes are the reference's "two
//    runner templates" — here two __device__ bodies sharing one LDS arena.
//  * decode items write their output directly (the in-item LDS merge
//    replaces the reference's reduction runner: no partials, no second
//    stage) — so the whole mixed batch is ONE launch with no epilogue.
//  * the host orders items most-expensive-first; the atomic ticket gives
//    dynamic load balance across CUs; outputs are per-item disjoint, so
//    execution order cannot change results (deterministic).
#include "fi/decode_mfma_body.hpp"
#include "fi/prefill_body.hpp"

namespace fi {

struct HolisticParams {
  PrefillParams pf;
  DecodeParams dec;
  const int32_t* items;      // [n_items][4]: kind, req, a, b (prefill first)
  int n_pf_items;            // items[0 .. n_pf_items) are prefill tiles
  int n_items;               // items[n_pf_items .. n_items) are decode items
  int n_dec_wgs;             // workgroups dedicated to the decode queue
  uint32_t* queue_head;      // [2] pf/dec tickets, zeroed before each launch
};

// KIND-PARTITIONED persistent loops: workgroups branch ONCE (uniformly, on
// blockIdx) into a prefill-only or decode-only queue loop. A single mixed
// loop was tried first — inlining both runners into one loop body made the
// register allocator spill ~500 B/lane (each body alone fits the 256-VGPR
// budget), and a noinline call split spilled MORE (ABI-reserved registers).
// Top-level disjoint loops keep each runner's allocation independent; the
// host sizes the decode pool by its cost fraction.
template <typename T, int HEAD_DIM, int GROUP_DEC, bool CAUSAL>
__global__ __launch_bounds__(512, 1) void batch_attention_kernel(HolisticParams h) {
  constexpr int PF_SMEM = prefill_tile_smem_bytes<T, HEAD_DIM, HEAD_DIM>();
  constexpr int DEC_SMEM =
      GROUP_DEC > 0 ? decode_mfma_smem_bytes<T, HEAD_DIM, GROUP_DEC>() : 0;
  __shared__ char smem[(PF_SMEM > DEC_SMEM ? PF_SMEM : DEC_SMEM)];
  __shared__ uint32_t s_item;
  auto drain_dec = [&]() {
    if constexpr (GROUP_DEC > 0) {
      for (;;) {
        __syncthreads();
        if (threadIdx.x == 0) s_item = atomicAdd(h.queue_head + 1, 1u);
        __syncthreads();
        uint32_t it = (uint32_t)h.n_pf_items + s_item;
        if (it >= (uint32_t)h.n_items) return;
        const int32_t* rec = h.items + 4 * it;
        // rec: (1, req, kv_head, qo_row = qo_indptr[req])
        decode_mfma_item_body<T, HEAD_DIM, GROUP_DEC>(h.dec, rec[1], rec[2], 0,
                                                      smem, rec[3]);
      }
    }
  };
  // decode-pool WGs drain their queue first, then steal prefill work;
  // prefill WGs do the reverse — no pool ever idles while work remains.
  if ((int)blockIdx.x < h.n_dec_wgs) drain_dec();
  for (;;) {
    __syncthreads();  // previous item's smem reads complete before re-claim
    if (threadIdx.x == 0) s_item = atomicAdd(h.queue_head, 1u);
    __syncthreads();
    uint32_t it = s_item;
    if (it >= (uint32_t)h.n_pf_items) break;
    const int32_t* rec = h.items + 4 * it;
    prefill_tile_body<T, T, HEAD_DIM, HEAD_DIM, 256, true, CAUSAL, false>(
        h.pf, rec[1], rec[2], rec[3], -1, smem);
  }
  if ((int)blockIdx.x >= h.n_dec_wgs) drain_dec();
}

template <typename T>
hipError_t holistic_dispatch(HolisticParams& h, int group_dec, bool causal,
                             int n_wgs, hipStream_t stream) {
  dim3 g((uint32_t)n_wgs), blk(512);
#define LAUNCH_H(HD, GD, CS) \
  hipLaunchKernelGGL((batch_attention_kernel<T, HD, GD, CS>), g, blk, 0, stream, h)
#define DISPATCH_HG(HD, CS)                               \
  do {                                                    \
    switch (group_dec) {                                  \
      case 0: LAUNCH_H(HD, 0, CS); break;                 \
      case 8: LAUNCH_H(HD, 8, CS); break;                 \
      default: return hipErrorInvalidValue;               \
    }                                                     \
  } while (0)
  int hd = h.pf.head_dim;
  if (hd == 128) { if (causal) DISPATCH_HG(128, true); else DISPATCH_HG(128, false); }
  else if (hd == 64) { if (causal) DISPATCH_HG(64, true); else DISPATCH_HG(64, false); }
  else return hipErrorInvalidValue;
#undef DISPATCH_HG
#undef LAUNCH_H
  return hipGetLastError();
}

}  // namespace fi

extern "C" hipError_t fi_batch_attention(int dtype, fi::HolisticParams* h,
                                         int group_dec, int causal, int n_wgs,
                                         hipStream_t stream) {
  if (h->n_items == 0) return hipSuccess;
  switch (dtype) {
    case 0: return fi::holistic_dispatch<fi::bf16>(*h, group_dec, causal != 0,
                                                   n_wgs, stream);
    case 1: return fi::holistic_dispatch<fi::fp16>(*h, group_dec, causal != 0,
                                                   n_wgs, stream);
  }
  return hipErrorInvalidValue;
}
