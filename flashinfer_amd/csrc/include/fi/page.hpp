// Paged KV-cache descriptor, layout-compatible with the reference's
// page-table contract (reference: include/flashinfer/page.cuh:38 paged_kv_t;
// docs/tutorials/kv_layout.rst): per-request page lists via
//   indptr[batch+1], indices[nnz_pages], last_page_len[batch]
// over a single KV buffer with layout
//   NHD: [max_pages, 2, page_size, num_kv_heads, head_dim]
//   HND: [max_pages, 2, num_kv_heads, page_size, head_dim]
// (the `2` axis is K=0 / V=1; separate K/V buffers are also supported by
// passing k_data/v_data with stride_page covering one page).
#pragma once
#include "fi/common.hpp"
#include "fi/fastdiv.hpp"

namespace fi {

enum class QKVLayout : int { kNHD = 0, kHND = 1 };

template <typename DType, typename IdType>
struct paged_kv_t {
  DType* k_data;   // base of K plane
  DType* v_data;   // base of V plane
  IdType* indices; // page indices  [nnz_pages]
  IdType* indptr;  // per-request page range [batch+1]
  IdType* last_page_len;  // valid entries in last page [batch]
  uint_fastdiv page_size;
  uint32_t num_heads;
  uint32_t head_dim;
  int64_t stride_page;  // elements between consecutive pages (same plane)
  int64_t stride_n;     // elements between consecutive tokens in a page
  int64_t stride_h;     // elements between heads in a page

  __host__ __device__ __forceinline__ int64_t kv_len(int batch_idx) const {
    int np = indptr[batch_idx + 1] - indptr[batch_idx];
    return np == 0 ? 0 : (int64_t)(np - 1) * page_size.d + last_page_len[batch_idx];
  }

  // element offset of (page_iter-th page of request, entry, head, feat)
  __device__ __forceinline__ int64_t get_elem_offset(int64_t page_id, uint32_t head,
                                                     uint32_t entry, uint32_t feat) const {
    return page_id * stride_page + head * stride_h + entry * stride_n + feat;
  }

  __device__ __forceinline__ DType* k_ptr(int64_t page_id, uint32_t head, uint32_t entry,
                                          uint32_t feat) const {
    return k_data + get_elem_offset(page_id, head, entry, feat);
  }
  __device__ __forceinline__ DType* v_ptr(int64_t page_id, uint32_t head, uint32_t entry,
                                          uint32_t feat) const {
    return v_data + get_elem_offset(page_id, head, entry, feat);
  }
};

}  // namespace fi
