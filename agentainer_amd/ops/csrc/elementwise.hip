// Fused elementwise kernels (gfx950) — memory-bound, bf16x8 vectorized.
//
// silu_mul: out = silu(gate) * up — the SwiGLU activation fused into one
// pass so the MLP never materializes silu(gate) (HBM3E round trips are the
// bound; cdna_hip_programming.md Appendix B "Element-wise").

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// gate/up may be strided row views (the two halves of the fused gate_up
// GEMM output); out is dense [T, H].
__global__ void silu_mul_kernel(short* __restrict__ out,
                                const short* __restrict__ gate,
                                const short* __restrict__ up, long n8,
                                int h8, long gs, long us) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n8) return;
  const long row = i / h8, col8 = i % h8;
  bf16x8 g = *reinterpret_cast<const bf16x8*>(gate + row * gs + col8 * 8);
  bf16x8 u = *reinterpret_cast<const bf16x8*>(up + row * us + col8 * 8);
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
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(gate.scalar_type() == at::kBFloat16, "silu_mul: bf16 only");
  TORCH_CHECK(gate.dim() == 2 && gate.stride(1) == 1 && up.stride(1) == 1);
  const int T = gate.size(0), H = gate.size(1);
  TORCH_CHECK(H % 8 == 0, "hidden must be a multiple of 8");
  const long n8 = (long)T * H / 8;
  const int block = 256;
  const long grid = (n8 + block - 1) / block;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(block), 0, stream,
                     (short*)out.data_ptr(), (const short*)gate.data_ptr(),
                     (const short*)up.data_ptr(), n8, H / 8,
                     (long)gate.stride(0), (long)up.stride(0));
}
