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

// Single-pass: the row lives in registers between the reduction and the
// scale, so x (and the fused residual sum) is read from HBM exactly once.
// 512 threads/block; VPT = ceil(H / 4096) bf16x8 vectors per thread.
template <bool FUSED_ADD, int VPT>
__global__ __launch_bounds__(512) void rmsnorm_kernel(
    short* __restrict__ out, const short* __restrict__ x,
    short* __restrict__ residual, const short* __restrict__ w, float eps,
    int H) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const long base = (long)row * H;
  __shared__ float red[8];

  bf16x8 v[VPT];
  float ss = 0.f;
#pragma unroll
  for (int r = 0; r < VPT; ++r) {
    const int i = (tid + r * 512) * 8;
    if (i < H) {
      v[r] = *reinterpret_cast<const bf16x8*>(x + base + i);
      if (FUSED_ADD) {
        bf16x8 rv = *reinterpret_cast<const bf16x8*>(residual + base + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[r][j] = f2bits(bits2f(v[r][j]) + bits2f(rv[j]));
        *reinterpret_cast<bf16x8*>(residual + base + i) = v[r];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bits2f(v[r][j]);
        ss = fmaf(f, f, ss);
      }
    } else {
      v[r] = bf16x8{};
    }
  }
  ss = wave_sum(ss);
  if ((tid & (WAVE - 1)) == 0) red[tid / WAVE] = ss;
  __syncthreads();
  ss = 0.f;
#pragma unroll
  for (int wv = 0; wv < 8; ++wv) ss += red[wv];
  const float rms = rsqrtf(ss / (float)H + eps);

#pragma unroll
  for (int r = 0; r < VPT; ++r) {
    const int i = (tid + r * 512) * 8;
    if (i < H) {
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
      bf16x8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = f2bits(bits2f(v[r][j]) * rms * bits2f(wv[j]));
      *reinterpret_cast<bf16x8*>(out + base + i) = ov;
    }
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
#define RN_LAUNCH(VPT)                                                         \
  hipLaunchKernelGGL((rmsnorm_kernel<false, VPT>), dim3(T), dim3(512), 0,      \
                     stream, (short*)out.data_ptr(),                           \
                     (const short*)x.data_ptr(), nullptr,                      \
                     (const short*)w.data_ptr(), (float)eps, H)
  if (H <= 4096) RN_LAUNCH(1);
  else if (H <= 8192) RN_LAUNCH(2);
  else if (H <= 16384) RN_LAUNCH(4);
  else TORCH_CHECK(false, "rmsnorm: hidden > 16384 unsupported");
#undef RN_LAUNCH
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
#define RNF_LAUNCH(VPT)                                                        \
  hipLaunchKernelGGL((rmsnorm_kernel<true, VPT>), dim3(T), dim3(512), 0,       \
                     stream, (short*)out.data_ptr(),                           \
                     (const short*)x.data_ptr(), (short*)residual.data_ptr(),  \
                     (const short*)w.data_ptr(), (float)eps, H)
  if (H <= 4096) RNF_LAUNCH(1);
  else if (H <= 8192) RNF_LAUNCH(2);
  else if (H <= 16384) RNF_LAUNCH(4);
  else TORCH_CHECK(false, "rmsnorm: hidden > 16384 unsupported");
#undef RNF_LAUNCH
}
