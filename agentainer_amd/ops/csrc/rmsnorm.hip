// RMSNorm kernels (gfx950) — memory-bound, bf16x8 vectorized.
//
// Serves the Llama/Mixtral forward of the agent engine (SURVEY.md §2.3
// "RMSNorm kernel" row; reference has no GPU code — this is new work).
//
// Two entry points:
//   rmsnorm(out, x, w, eps):            out = x * rsqrt(mean(x^2)+eps) * w
//   fused_add_rmsnorm(out, x, res, w):  res += x;  out = rmsnorm(res)
//
// Design notes (cdna_hip_programming.md G13): hipcc does not vectorize
// scalar bf16 loads — all global access is 16 B/lane (bf16x8). One
// workgroup (256 threads) per row; f32 accumulation; wave shuffle + LDS
// cross-wave reduction. Hidden sizes up to 16384 handled by grid-stride
// over 2048-element chunks.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(short* __restrict__ out,
                               const short* __restrict__ x,
                               short* __restrict__ residual,
                               const short* __restrict__ w, float eps, int H) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const long base = (long)row * H;
  __shared__ float red[4];

  // pass 1: sum of squares (and the fused residual add, kept in registers
  // only when H fits one pass; otherwise re-read)
  float ss = 0.f;
  for (int i = tid * 8; i < H; i += blockDim.x * 8) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + base + i);
    if (FUSED_ADD) {
      bf16x8 rv = *reinterpret_cast<const bf16x8*>(residual + base + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = bits2f(xv[j]) + bits2f(rv[j]);
        xv[j] = f2bits(s);
      }
      *reinterpret_cast<bf16x8*>(residual + base + i) = xv;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bits2f(xv[j]);
      ss = fmaf(f, f, ss);
    }
  }
  ss = wave_sum(ss);
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) red[wid] = ss;
  __syncthreads();
  ss = red[0] + red[1] + red[2] + red[3];
  const float rms = rsqrtf(ss / (float)H + eps);

  // pass 2: scale+weight (reads the post-add residual when fused)
  const short* src = FUSED_ADD ? residual : x;
  for (int i = tid * 8; i < H; i += blockDim.x * 8) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(src + base + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov[j] = f2bits(bits2f(xv[j]) * rms * bits2f(wv[j]));
    *reinterpret_cast<bf16x8*>(out + base + i) = ov;
  }
}

}  // namespace

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "rmsnorm: bf16 only");
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  const int T = x.numel() / H;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((rmsnorm_kernel<false>), dim3(T), dim3(256), 0, stream,
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     nullptr, (const short*)w.data_ptr(), (float)eps, H);
}

void fused_add_rmsnorm(torch::Tensor out, torch::Tensor x,
                       torch::Tensor residual, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous() && w.is_contiguous() &&
              residual.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "fused_add_rmsnorm: bf16 only");
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  const int T = x.numel() / H;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((rmsnorm_kernel<true>), dim3(T), dim3(256), 0, stream,
                     (short*)out.data_ptr(), (const short*)x.data_ptr(),
                     (short*)residual.data_ptr(), (const short*)w.data_ptr(),
                     (float)eps, H);
}
