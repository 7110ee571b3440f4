// Fused elementwise kernels (gfx950) — memory-bound, bf16x8 vectorized.
//
// silu_mul: out = silu(gate) * up — the SwiGLU activation fused into one
// pass so the MLP never materializes silu(gate) (HBM3E round trips are the
// bound; cdna_hip_programming.md Appendix B "Element-wise").

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

__global__ void silu_mul_kernel(short* __restrict__ out,
                                const short* __restrict__ gate,
                                const short* __restrict__ up, long n8) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n8) return;
  bf16x8 g = reinterpret_cast<const bf16x8*>(gate)[i];
  bf16x8 u = reinterpret_cast<const bf16x8*>(up)[i];
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float x = bits2f(g[j]);
    float s = x / (1.0f + __expf(-x));
    o[j] = f2bits(s * bits2f(u[j]));
  }
  reinterpret_cast<bf16x8*>(out)[i] = o;
}

}  // namespace

void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.is_contiguous() && up.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(gate.scalar_type() == at::kBFloat16, "silu_mul: bf16 only");
  TORCH_CHECK(gate.numel() % 8 == 0, "numel must be a multiple of 8");
  const long n8 = gate.numel() / 8;
  const int block = 256;
  const long grid = (n8 + block - 1) / block;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(block), 0, stream,
                     (short*)out.data_ptr(), (const short*)gate.data_ptr(),
                     (const short*)up.data_ptr(), n8);
}
