// Empirical probe for gfx950 mfma_scale_f32_16x16x128_f8f6f4 operand
// layouts (the scaled-MX instruction's lane maps are not in the shipped
// docs). Hypothesis under test (natural extension of the 16x16x32 maps):
//   A: lane l holds A[l%16][(l/16)*32 + j], j = 0..31 (one MX block)
//   B: lane l holds B[(l/16)*32 + j][l%16]
//   C: lane l holds rows (l/16)*4 + r, col l%16
//   scale ints: low byte = e8m0 scale of the lane's 32-elem block
// Builtin arg order (LLVM): (a, b, c, cbsz=fmtA, blgp=fmtB, opsel_a,
// scale_a, opsel_b, scale_b); fmt codes 0=fp8e4m3, 4=fp4.
// Driver: tools/mx_probe.py compares against a host fp32 reference.
#include <hip/hip_runtime.h>
#include <cstdint>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef int i32x8 __attribute__((ext_vector_type(8)));

extern "C" __global__ void mx_gemm_16x16x128(
    float* __restrict__ c_out,          // [16, 16] row-major
    const unsigned char* __restrict__ a_frag,  // [64, 32] per-lane bytes
    const unsigned char* __restrict__ b_frag,  // [64, 32]
    const int* __restrict__ sa,         // [64]
    const int* __restrict__ sb,         // [64]
    int fmt) {
  const int lane = threadIdx.x;
  i32x8 a = *reinterpret_cast<const i32x8*>(a_frag + lane * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(b_frag + lane * 32);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  if (fmt == 0) {
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 0, 0, sa[lane], 0, sb[lane]);
  } else if (fmt == 4) {
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 4, 4, 0, sa[lane], 0, sb[lane]);
  } else if (fmt == 10) {  // mixed: A fp8, B fp4 via (cbsz=0, blgp=4)
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 4, 0, sa[lane], 0, sb[lane]);
  } else if (fmt == 11) {  // mixed: A fp8, B fp4 via (cbsz=4, blgp=0)
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 4, 0, 0, sa[lane], 0, sb[lane]);
  }
  const int col = lane % 16;
  const int r0 = (lane / 16) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) c_out[(r0 + r) * 16 + col] = acc[r];
}

extern "C" void mx_probe_launch(float* c_out, const unsigned char* a_frag,
                                const unsigned char* b_frag, const int* sa,
                                const int* sb, int fmt) {
  hipLaunchKernelGGL(mx_gemm_16x16x128, dim3(1), dim3(64), 0, 0, c_out,
                     a_frag, b_frag, sa, sb, fmt);
  hipDeviceSynchronize();
}
