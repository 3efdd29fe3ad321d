// Paged KV-cache maintenance kernels. Parity with reference
// include/flashinfer/page.cuh (AppendPagedKVCacheKernel:395,
// get_batch_indices_positions flashinfer/page.py:251).
#include "fi/common.hpp"
#include "fi/page.hpp"
#include "fi/vec.hpp"

namespace fi {

// Scatter nnz append tokens into the paged cache. TC is the cache element
// type; when TC == fp8 the values are quantized on the fly with the given
// per-tensor scales (reference fp8/NVFP4 quantize-append role).
// k/v append buffers: [nnz, num_kv_heads, head_dim] (row-major strides given).
template <typename T, typename TC, typename IdType, int VEC>
__global__ void append_paged_kv_cache_kernel(paged_kv_t<TC, IdType> paged,
                                             const T* __restrict__ k,
                                             const T* __restrict__ v,
                                             const IdType* __restrict__ batch_indices,
                                             const IdType* __restrict__ positions,
                                             int64_t nnz, int64_t k_stride_n,
                                             int64_t k_stride_h, int64_t v_stride_n,
                                             int64_t v_stride_h, float inv_k_scale,
                                             float inv_v_scale) {
  int H = paged.num_heads, D = paged.head_dim;
  int chunks = D / VEC;
  int64_t total = nnz * H * chunks;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(idx % chunks);
    int64_t th = idx / chunks;
    int h = (int)(th % H);
    int64_t i = th / H;
    uint32_t b = (uint32_t)batch_indices[i];
    uint32_t pos = (uint32_t)positions[i];
    uint32_t page_iter, entry;
    paged.page_size.divmod(pos, page_iter, entry);
    int64_t page_id = paged.indices[paged.indptr[b] + page_iter];
    int64_t off = paged.get_elem_offset(page_id, h, entry, c * VEC);
    vec_t<T, VEC> kv, vv;
    kv.load(k + i * k_stride_n + h * k_stride_h + c * VEC);
    vv.load(v + i * v_stride_n + h * v_stride_h + c * VEC);
    if constexpr (__is_same(T, TC)) {
      kv.store(paged.k_data + off);
      vv.store(paged.v_data + off);
    } else {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        paged.k_data[off + j] = TC(kv.get(j) * inv_k_scale);
        paged.v_data[off + j] = TC(vv.get(j) * inv_v_scale);
      }
    }
  }
}

template <typename IdType>
__global__ void batch_indices_positions_kernel(const IdType* __restrict__ append_indptr,
                                               const IdType* __restrict__ seq_lens,
                                               IdType* __restrict__ batch_indices,
                                               IdType* __restrict__ positions, int batch) {
  int b = blockIdx.x;
  if (b >= batch) return;
  IdType start = append_indptr[b], end = append_indptr[b + 1];
  IdType append_len = end - start;
  for (IdType i = threadIdx.x; i < append_len; i += blockDim.x) {
    batch_indices[start + i] = b;
    positions[start + i] = seq_lens[b] - append_len + i;
  }
}

}  // namespace fi

extern "C" hipError_t fi_append_paged_kv_cache(
    int dtype, int cache_dtype, void* k_data, void* v_data, const int32_t* indices,
    const int32_t* indptr, const int32_t* last_page_len, int page_size, int num_heads,
    int head_dim, int64_t stride_page, int64_t stride_n, int64_t stride_h, const void* k,
    const void* v, const int32_t* batch_indices, const int32_t* positions, int64_t nnz,
    int64_t k_stride_n, int64_t k_stride_h, int64_t v_stride_n, int64_t v_stride_h,
    float k_scale, float v_scale, hipStream_t stream) {
  int vec = (head_dim % 8 == 0) ? 8 : 1;
  int64_t total = nnz * num_heads * (head_dim / vec);
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
  if (grid == 0) grid = 1;

#define LAUNCH_P(T, TC, VEC)                                                          \
  do {                                                                                \
    fi::paged_kv_t<TC, int32_t> paged;                                                \
    paged.k_data = (TC*)k_data;                                                       \
    paged.v_data = (TC*)v_data;                                                       \
    paged.indices = (int32_t*)indices;                                                \
    paged.indptr = (int32_t*)indptr;                                                  \
    paged.last_page_len = (int32_t*)last_page_len;                                    \
    paged.page_size = fi::uint_fastdiv(page_size);                                    \
    paged.num_heads = num_heads;                                                      \
    paged.head_dim = head_dim;                                                        \
    paged.stride_page = stride_page;                                                  \
    paged.stride_n = stride_n;                                                        \
    paged.stride_h = stride_h;                                                        \
    hipLaunchKernelGGL((fi::append_paged_kv_cache_kernel<T, TC, int32_t, VEC>),      \
                       dim3(grid), dim3(256), 0, stream, paged, (const T*)k,          \
                       (const T*)v, batch_indices, positions, nnz, k_stride_n,        \
                       k_stride_h, v_stride_n, v_stride_h, 1.f / k_scale,             \
                       1.f / v_scale);                                                \
  } while (0)

  if (cache_dtype == dtype) {
    switch (dtype) {
      case 0:
        if (vec == 8) LAUNCH_P(fi::bf16, fi::bf16, 8);
        else LAUNCH_P(fi::bf16, fi::bf16, 1);
        break;
      case 1:
        if (vec == 8) LAUNCH_P(fi::fp16, fi::fp16, 8);
        else LAUNCH_P(fi::fp16, fi::fp16, 1);
        break;
      case 2:
        if (vec == 8) LAUNCH_P(float, float, 8);
        else LAUNCH_P(float, float, 1);
        break;
      default:
        return hipErrorInvalidValue;
    }
  } else if (cache_dtype == 3 && dtype == 0) {
    if (vec == 8) LAUNCH_P(fi::bf16, fi::fp8_e4m3, 8);
    else LAUNCH_P(fi::bf16, fi::fp8_e4m3, 1);
  } else if (cache_dtype == 3 && dtype == 1) {
    if (vec == 8) LAUNCH_P(fi::fp16, fi::fp8_e4m3, 8);
    else LAUNCH_P(fi::fp16, fi::fp8_e4m3, 1);
  } else {
    return hipErrorInvalidValue;
  }
#undef LAUNCH_P
  return hipGetLastError();
}

extern "C" hipError_t fi_batch_indices_positions(const int32_t* append_indptr,
                                                 const int32_t* seq_lens,
                                                 int32_t* batch_indices,
                                                 int32_t* positions, int batch,
                                                 hipStream_t stream) {
  if (batch == 0) return hipSuccess;
  hipLaunchKernelGGL((fi::batch_indices_positions_kernel<int32_t>), dim3(batch), dim3(256),
                     0, stream, append_indptr, seq_lens, batch_indices, positions, batch);
  return hipGetLastError();
}
