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
