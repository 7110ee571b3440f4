// Rotary position embedding (gfx950) — applied in place to Q and K.
//
// NeoX/Llama rotation of pairs (x[i], x[i+D/2]) with a HOST-precomputed
// f32 cos/sin table (on-device sinf/cosf turns a memory-bound op
// VALU-bound — cdna_hip_programming.md Appendix B "trig-heavy ops").
//
// cos_sin: [max_pos, D] f32 laid out as [cos(0..D/2) | sin(0..D/2)] per row.
// q: [T, n_q, D] bf16; k: [T, n_kv, D] bf16; positions: [T] int32.
// Serves the Llama/Mixtral forward (SURVEY.md §2.3 "RoPE kernel" row).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// grid: (T, n_q + n_kv); block: D/2 lanes (<=256). Each lane rotates one
// (x1, x2) pair; each pair of bf16 loads is 2 B, cos/sin are two f32x1
// loads from a row the whole block shares (L1-resident).
// q/k may be strided views into the fused qkv GEMM output (token stride
// qs/ks) — avoids materializing contiguous q/k copies per layer.
__global__ void rope_kernel(short* __restrict__ q, short* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int* __restrict__ positions, int n_q,
                            int n_kv, int D, long qs, long ks) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;  // pair index < D/2
  const int half = D / 2;
  if (i >= half) return;
  short* base = (h < n_q) ? q + (long)t * qs + (long)h * D
                          : k + (long)t * ks + (long)(h - n_q) * D;
  const int pos = positions[t];
  const float c = cos_sin[(long)pos * D + i];
  const float s = cos_sin[(long)pos * D + half + i];
  const float x1 = bits2f(base[i]);
  const float x2 = bits2f(base[half + i]);
  base[i] = f2bits(x1 * c - x2 * s);
  base[half + i] = f2bits(x1 * s + x2 * c);
}

}  // namespace

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
                  torch::Tensor positions) {
  TORCH_CHECK(cos_sin.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16 && k.scalar_type() == at::kBFloat16);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  const int T = q.size(0), n_q = q.size(1), D = q.size(2);
  const int n_kv = k.size(1);
  TORCH_CHECK(k.size(0) == T && k.size(2) == D);
  TORCH_CHECK(D % 2 == 0 && D / 2 <= 1024);
  // strided views allowed: last dim contiguous, head stride == D
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D);
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == D);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_kernel, dim3(T, n_q + n_kv), dim3(D / 2), 0, stream,
                     (short*)q.data_ptr(), (short*)k.data_ptr(),
                     cos_sin.data_ptr<float>(), positions.data_ptr<int>(),
                     n_q, n_kv, D, (long)q.stride(0), (long)k.stride(0));
}

// ---------------------------------------------------------------------------
// Fused RoPE + KV-append: rotates q in place, rotates k in registers and
// scatters k/v straight into the paged pools — one kernel instead of two,
// and k never takes the rotate-write-reload round trip through HBM.
// grid: (T, n_q + n_kv); blocks for h < n_q rotate q rows; blocks for
// kv heads rotate k and append k+v.

namespace {

// CT = short (bf16 cache) or unsigned char (fp8 e4m3 cache)
template <typename CT>
__device__ __forceinline__ CT f2cache(float x);
template <>
__device__ __forceinline__ short f2cache<short>(float x) { return f2bits(x); }
template <>
__device__ __forceinline__ unsigned char f2cache<unsigned char>(float x) {
  return f2fp8(x);
}

template <typename CT>
__global__ void rope_append_kernel(
    short* __restrict__ q, short* __restrict__ k, const short* __restrict__ v,
    CT* __restrict__ k_cache, CT* __restrict__ v_cache,
    const float* __restrict__ cos_sin, const int* __restrict__ positions,
    const long* __restrict__ slots, int n_q, int n_kv, int D, int PS,
    long qs, long ks, long vs) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;  // pair index < D/2
  const int half = D / 2;
  if (i >= half) return;
  const int pos = positions[t];
  const float c = cos_sin[(long)pos * D + i];
  const float s = cos_sin[(long)pos * D + half + i];
  if (h < n_q) {
    short* base = q + (long)t * qs + (long)h * D;
    const float x1 = bits2f(base[i]);
    const float x2 = bits2f(base[half + i]);
    base[i] = f2bits(x1 * c - x2 * s);
    base[half + i] = f2bits(x1 * s + x2 * c);
    return;
  }
  const int g = h - n_q;
  const long slot = slots[t];
  const short* kb = k + (long)t * ks + (long)g * D;
  const float x1 = bits2f(kb[i]);
  const float x2 = bits2f(kb[half + i]);
  const CT r1 = f2cache<CT>(x1 * c - x2 * s);
  const CT r2 = f2cache<CT>(x1 * s + x2 * c);
  if (slot < 0) return;
  const long page = slot / PS;
  const int off = (int)(slot % PS);
  const int D8 = D / 8;
  // k_cache[page][g][d8][off][j]
  long kbase = (((page * n_kv + g) * D8) * PS + off) * 8;
  k_cache[kbase + (i / 8) * PS * 8 + i % 8] = r1;
  k_cache[kbase + ((half + i) / 8) * PS * 8 + (half + i) % 8] = r2;
  // v rows are not rotated: copy 2 elements per lane
  const short* vb = v + (long)t * vs + (long)g * D;
  long vbase = (((page * n_kv + g) * PS) + off) * D;
  v_cache[vbase + i] = f2cache<CT>(bits2f(vb[i]));
  v_cache[vbase + half + i] = f2cache<CT>(bits2f(vb[half + i]));
}

}  // namespace

void rope_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor k_cache, torch::Tensor v_cache,
                 torch::Tensor cos_sin, torch::Tensor positions,
                 torch::Tensor slot_mapping) {
  TORCH_CHECK(cos_sin.is_contiguous() && cos_sin.scalar_type() == at::kFloat);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T = q.size(0), n_q = q.size(1), D = q.size(2);
  const int n_kv = k.size(1), PS = k_cache.size(3);
  TORCH_CHECK(D % 16 == 0 && D / 2 <= 1024);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D);
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == D);
  TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == D);
  if (T == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  if (k_cache.scalar_type() == at::kFloat8_e4m3fn) {
    hipLaunchKernelGGL(rope_append_kernel<unsigned char>,
                       dim3(T, n_q + n_kv), dim3(D / 2), 0, stream,
                       (short*)q.data_ptr(), (short*)k.data_ptr(),
                       (const short*)v.data_ptr(),
                       (unsigned char*)k_cache.data_ptr(),
                       (unsigned char*)v_cache.data_ptr(),
                       cos_sin.data_ptr<float>(), positions.data_ptr<int>(),
                       slot_mapping.data_ptr<long>(), n_q, n_kv, D, PS,
                       (long)q.stride(0), (long)k.stride(0),
                       (long)v.stride(0));
    return;
  }
  TORCH_CHECK(k_cache.scalar_type() == at::kBFloat16);
  hipLaunchKernelGGL(rope_append_kernel<short>, dim3(T, n_q + n_kv),
                     dim3(D / 2), 0,
                     stream, (short*)q.data_ptr(), (short*)k.data_ptr(),
                     (const short*)v.data_ptr(), (short*)k_cache.data_ptr(),
                     (short*)v_cache.data_ptr(), cos_sin.data_ptr<float>(),
                     positions.data_ptr<int>(), slot_mapping.data_ptr<long>(),
                     n_q, n_kv, D, PS, (long)q.stride(0), (long)k.stride(0),
                     (long)v.stride(0));
}
