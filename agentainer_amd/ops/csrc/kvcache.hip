// Paged KV-cache kernels (gfx950).
//
// The KV pool replaces the reference's "container keeps its filesystem"
// durability (SURVEY.md §2.3 "KV-cache manager" row): each agent's
// conversation lives as pages in HBM3E; stop/checkpoint gathers pages into
// a contiguous buffer streamed to pinned host memory via hipMemcpyAsync.
//
// Layouts, chosen for the decode kernel's coalescing (decode_attn.hip):
//   k_cache: [n_pages, n_kv, D/8, page_size, 8]  bf16  ("x=8": lanes over
//            tokens read 16 B contiguous per (d8, token) pair)
//   v_cache: [n_pages, n_kv, page_size, D]       bf16  (natural rows; lanes
//            over dims read contiguous)
//
// kv_append scatters the current step's K/V (post-RoPE) into the pools by
// slot id = page * page_size + offset; slot -1 = skip (padded token).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// grid: (T); block: 256. Each thread moves 16 B; loops over n_kv * D/8
// chunks for K and n_kv * D/8 chunks for V.
// CT = short (bf16 pool) or unsigned char (fp8 e4m3 pool)
template <typename CT>
__device__ __forceinline__ void store8(CT* dst, bf16x8 src);
template <>
__device__ __forceinline__ void store8<short>(short* dst, bf16x8 src) {
  *reinterpret_cast<bf16x8*>(dst) = src;
}
template <>
__device__ __forceinline__ void store8<unsigned char>(unsigned char* dst,
                                                      bf16x8 src) {
  u8x8 b;
#pragma unroll
  for (int j = 0; j < 8; ++j) b[j] = f2fp8(bits2f(src[j]));
  *reinterpret_cast<u8x8*>(dst) = b;
}

template <typename CT>
__global__ void kv_append_kernel(CT* __restrict__ k_cache,
                                 CT* __restrict__ v_cache,
                                 const short* __restrict__ k,
                                 const short* __restrict__ v,
                                 const long* __restrict__ slots, int n_kv,
                                 int D, int PS, long kts, long vts) {
  const int t = blockIdx.x;
  const long slot = slots[t];
  if (slot < 0) return;
  const long page = slot / PS;
  const int off = (int)(slot % PS);
  const int D8 = D / 8;
  const int chunks = n_kv * D8;
  for (int c = threadIdx.x; c < chunks; c += blockDim.x) {
    const int h = c / D8, d8 = c % D8;
    bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(k + (long)t * kts + (long)h * D + d8 * 8);
    // k_cache[page][h][d8][off][0..8]
    long kidx = ((((page * n_kv + h) * D8 + d8) * PS) + off) * 8;
    store8<CT>(k_cache + kidx, kv8);
    bf16x8 vv8 = *reinterpret_cast<const bf16x8*>(v + (long)t * vts + (long)h * D + d8 * 8);
    // v_cache[page][h][off][d8*8..]
    long vidx = (((page * n_kv + h) * PS) + off) * D + d8 * 8;
    store8<CT>(v_cache + vidx, vv8);
  }
}

// Gather/scatter whole pages between the pools and a contiguous buffer
// dst/src: [n_pages_sel, bytes_per_page/2] viewed as shorts, where one
// page's image is [K plane | V plane] for all kv heads.
__global__ void page_copy_kernel(short* __restrict__ k_cache,
                                 short* __restrict__ v_cache,
                                 short* __restrict__ buf,
                                 const int* __restrict__ page_ids, long plane,
                                 bool to_buf) {
  const int p = blockIdx.x;              // index into page_ids
  const long page = page_ids[p];
  short* kp = k_cache + page * plane;
  short* vp = v_cache + page * plane;
  short* bk = buf + (long)p * 2 * plane;
  short* bv = bk + plane;
  for (long i = (long)threadIdx.x * 8; i < plane; i += (long)blockDim.x * 8) {
    if (to_buf) {
      *reinterpret_cast<bf16x8*>(bk + i) = *reinterpret_cast<bf16x8*>(kp + i);
      *reinterpret_cast<bf16x8*>(bv + i) = *reinterpret_cast<bf16x8*>(vp + i);
    } else {
      *reinterpret_cast<bf16x8*>(kp + i) = *reinterpret_cast<bf16x8*>(bk + i);
      *reinterpret_cast<bf16x8*>(vp + i) = *reinterpret_cast<bf16x8*>(bv + i);
    }
  }
}

}  // namespace

void kv_append(torch::Tensor k_cache, torch::Tensor v_cache, torch::Tensor k,
               torch::Tensor v, torch::Tensor slot_mapping) {
  TORCH_CHECK(k_cache.scalar_type() == at::kBFloat16 ||
              k_cache.scalar_type() == at::kFloat8_e4m3fn);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T = k.size(0), n_kv = k.size(1), D = k.size(2);
  const int PS = k_cache.size(3);
  TORCH_CHECK(k_cache.size(1) == n_kv && k_cache.size(2) == D / 8);
  TORCH_CHECK(v_cache.size(2) == PS && v_cache.size(3) == D);
  TORCH_CHECK(D % 8 == 0);
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == D);
  TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == D);
  if (T == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  if (k_cache.scalar_type() == at::kFloat8_e4m3fn) {
    hipLaunchKernelGGL(kv_append_kernel<unsigned char>, dim3(T), dim3(256), 0,
                       stream, (unsigned char*)k_cache.data_ptr(),
                       (unsigned char*)v_cache.data_ptr(),
                       (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                       slot_mapping.data_ptr<long>(), n_kv, D, PS,
                       (long)k.stride(0), (long)v.stride(0));
    return;
  }
  hipLaunchKernelGGL(kv_append_kernel<short>, dim3(T), dim3(256), 0, stream,
                     (short*)k_cache.data_ptr(), (short*)v_cache.data_ptr(),
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     slot_mapping.data_ptr<long>(), n_kv, D, PS,
                     (long)k.stride(0), (long)v.stride(0));
}

static void page_copy(torch::Tensor k_cache, torch::Tensor v_cache,
                      torch::Tensor buf, torch::Tensor page_ids, bool to_buf) {
  TORCH_CHECK(page_ids.scalar_type() == at::kInt);
  const int n_kv = k_cache.size(1), D = k_cache.size(2) * 8, PS = k_cache.size(3);
  const int n = page_ids.size(0);
  if (n == 0) return;
  TORCH_CHECK(buf.numel() >= (long)n * 2 * n_kv * D * PS,
              "page buffer too small");
  TORCH_CHECK(buf.element_size() == k_cache.element_size());
  // raw byte copy expressed in shorts: plane bytes are even for both bf16
  // (2 B/elem) and fp8 (1 B/elem, D=128 keeps it 16 B aligned)
  const long plane = (long)n_kv * D * PS * k_cache.element_size() / 2;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(page_copy_kernel, dim3(n), dim3(256), 0, stream,
                     (short*)k_cache.data_ptr(), (short*)v_cache.data_ptr(),
                     (short*)buf.data_ptr(), page_ids.data_ptr<int>(), plane,
                     to_buf);
}

void gather_kv_pages(torch::Tensor dst, torch::Tensor k_cache,
                     torch::Tensor v_cache, torch::Tensor page_ids) {
  page_copy(k_cache, v_cache, dst, page_ids, /*to_buf=*/true);
}

void scatter_kv_pages(torch::Tensor k_cache, torch::Tensor v_cache,
                      torch::Tensor src, torch::Tensor page_ids) {
  page_copy(k_cache, v_cache, src, page_ids, /*to_buf=*/false);
}
