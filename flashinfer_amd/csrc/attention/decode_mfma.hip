// MFMA fused decode — the GQA small-batch latency shape (GROUP >= 8).
//
// Role parity: reference decode.cuh:615 + the "tensor-core decode" design of
// flashinfer/decode.py:1697, but a dedicated CDNA4 kernel instead of reusing
// the 128-row prefill tile: at bs=16 the prefill reuse runs 128-row MFMA
// tiles that are 94% padding AND underfills the grid; the vector kernel burns
// 8 dot+shfl chains per K read (ds_bpermute-bound at GQA-8, measured 78 us at
// the BASELINE bs=16/kv=1024 config). Here:
//
//  * one workgroup per (request, kv_head); 4 waves each own a contiguous
//    quarter of the request's KV — NO __syncthreads in the main loop (each
//    wave stages into its own LDS region; LDS ops of one wave complete in
//    issue order, so single-buffered stage + register ring is race-free).
//  * per wave: 32-token KV tiles, S^T = mfma32(K_frag, Q_frag) with the GQA
//    group (8/16/32 q heads) as the 32-column q dimension — one MFMA pass
//    amortizes the whole group; decode q rows are all at the same position so
//    bounds/window/soft-cap/ALiBi masks are lane-uniform or per-lane scalars.
//  * same softmax-in-registers dance as the prefill kernel (lane owns one q
//    column, defer-max, v_cvt_pk_bf16_f32 P-pack, lazy half-exchange,
//    ds_read_b64_tr_b16 V^T fragments).
//  * epilogue: the 4 per-wave online-softmax states merge through LDS (one
//    barrier in the whole kernel) and the output is written directly in the
//    q dtype — one launch, no f32 round-trip, no merge kernel.
#include "fi/decode_mfma_body.hpp"

namespace fi {

template <typename T, int HEAD_DIM, int GROUP, typename TKV = T>
__global__ __launch_bounds__(kDecWaves * 64, 1) void decode_mfma_kernel(DecodeParams p) {
  __shared__ char smem[decode_mfma_smem_bytes<T, HEAD_DIM, GROUP>()];
  decode_mfma_item_body<T, HEAD_DIM, GROUP, TKV>(p, blockIdx.x, blockIdx.y,
                                                 blockIdx.z, smem);
}

template <typename T, typename TKV = T>
hipError_t decode_mfma_dispatch(DecodeParams& p, hipStream_t stream) {
  int group = p.num_qo_heads / p.num_kv_heads;
  dim3 g((uint32_t)p.batch, (uint32_t)p.num_kv_heads,
         (uint32_t)(p.split > 1 ? p.split : 1));
  dim3 blk(kDecWaves * 64);
#define LAUNCH_M(HD, G) \
  hipLaunchKernelGGL((decode_mfma_kernel<T, HD, G, TKV>), g, blk, 0, stream, p)
  // every group <= 32 rides the same 32x32 tile: the q dim is padding-
  // tolerant (QROWS rows live, rest zeroed), so the QK/PV cost per KV tile
  // is identical for GROUP 1..32 — small groups (MHA, GQA-2/4, Mixtral's 6,
  // Yi's 7) get the MFMA pipeline instead of the scalar vector fallback.
#define LAUNCH_SW(HD)                                 \
    switch (group) {                                  \
      case 1: LAUNCH_M(HD, 1); break;                 \
      case 2: LAUNCH_M(HD, 2); break;                 \
      case 4: LAUNCH_M(HD, 4); break;                 \
      case 5: LAUNCH_M(HD, 5); break;                 \
      case 6: LAUNCH_M(HD, 6); break;                 \
      case 7: LAUNCH_M(HD, 7); break;                 \
      case 8: LAUNCH_M(HD, 8); break;                 \
      case 16: LAUNCH_M(HD, 16); break;               \
      case 32: LAUNCH_M(HD, 32); break;               \
      default: return hipErrorInvalidValue;           \
    }
  if (p.head_dim == 128) {
    LAUNCH_SW(128);
  } else if (p.head_dim == 64) {
    LAUNCH_SW(64);
  } else {
    return hipErrorInvalidValue;
  }
#undef LAUNCH_SW
#undef LAUNCH_M
  return hipGetLastError();
}

}  // namespace fi

extern "C" hipError_t fi_decode_mfma(int dtype, int kv_dtype, fi::DecodeParams* p,
                                     hipStream_t stream) {
  if (p->batch == 0) return hipSuccess;
  if (kv_dtype == dtype) {
    switch (dtype) {
      case 0: return fi::decode_mfma_dispatch<fi::bf16>(*p, stream);
      case 1: return fi::decode_mfma_dispatch<fi::fp16>(*p, stream);
    }
  } else if (kv_dtype == 3 && dtype == 0) {
    return fi::decode_mfma_dispatch<fi::bf16, fi::fp8_e4m3>(*p, stream);
  } else if (kv_dtype == 3 && dtype == 1) {
    return fi::decode_mfma_dispatch<fi::fp16, fi::fp8_e4m3>(*p, stream);
  }
  return hipErrorInvalidValue;
}
