// Skinny-M GEMM (gfx950, MFMA bf16) — the decode-step projection GEMM.
//
// out[M,N] = x[M,K] @ W[N,K]^T for M <= 64 (torch Linear weight layout).
// At decode batch sizes hipBLASLt reaches only ~2-4.5 TB/s of weight
// streaming on these shapes (profiled); this kernel exists to stream W at
// the HBM roofline:
//
//   * W rows feed MFMA B-fragments STRAIGHT from HBM: B lane l holds
//     W[n0 + l%16][k + (l/16)*8 + j] — a single contiguous 16 B load per
//     fragment, no LDS staging (cdna_hip_programming.md §5 "GEMV / M<=16
//     decode weights: load straight to VGPRs" — generalized to MFMA).
//   * x (<=512 KB) is L2-resident and re-read by every n-tile block —
//     A-fragments also load straight from global.
//   * Grid: (N/64) x SPLITK blocks of 4 waves; each wave owns 16 n-cols
//     and 4 m-sub accumulators (16x16x32 MFMA). Split-K keeps >=256
//     blocks in flight for small N; fp32 partials combine in a second
//     elementwise kernel.
//
// Used by the engine for decode projections; prefill (large M) stays on
// hipBLASLt (compute-bound regime where the library is at ~1.2 PF).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __bf16 bf16v8 __attribute__((ext_vector_type(8)));

template <bool SPLIT, bool FULL>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    void* __restrict__ out,            // bf16 [M,N] or f32 ws [S,M,N]
    const short* __restrict__ x,       // [M,K]
    const short* __restrict__ w,       // [N,K]
    int M, int N, int K, int k_per_split) {
  const int n0 = blockIdx.x * 64 + (threadIdx.x / WAVE) * 16;
  const int split = blockIdx.y;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane % 16;
  const int lg = lane / 16;
  const int k0 = split * k_per_split;
  const int k1 = min(k0 + k_per_split, K);

  f32x4 acc[4];
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) acc[ms] = f32x4{0.f, 0.f, 0.f, 0.f};

  const short* wp = w + (long)(n0 + l16) * K;

  // x tile [64][KC] staged through LDS once per BLOCK per k-chunk — the
  // 4 waves reading x straight from global exceeded the per-CU L2
  // bandwidth (x re-read once per wave instead of once per block).
  // Double-buffered, T14 split: issue next tile's loads early, ds_write
  // after the MFMAs (guide §6 G15). Row stride 72 shorts (144 B) makes the
  // b128 fragment reads conflict-free (row*36 mod 64 distinct over 16).
  constexpr int KC = 64;
  constexpr int XS = KC + 8;
  __shared__ short x_lds[2][64 * XS];
  const int s_row = threadIdx.x % 64;        // staging: this thread's x row
  const int s_col8 = threadIdx.x / 64;       // covers cols {0..3}*8, +32 next
  const short* s_xp = x + (long)(FULL ? s_row : min(s_row, M - 1)) * K;
  const bool s_alive = FULL || s_row < M;

  bf16x8 st0{}, st1{};
  // prologue: stage first tile
  if (s_alive) {
    st0 = *reinterpret_cast<const bf16x8*>(s_xp + k0 + s_col8 * 8);
    st1 = *reinterpret_cast<const bf16x8*>(s_xp + k0 + 32 + s_col8 * 8);
  }
  *reinterpret_cast<bf16x8*>(&x_lds[0][s_row * XS + s_col8 * 8]) = st0;
  *reinterpret_cast<bf16x8*>(&x_lds[0][s_row * XS + 32 + s_col8 * 8]) = st1;
  __syncthreads();

  int buf = 0;
  for (int k = k0; k < k1; k += KC) {
    const bool has_next = (k + KC < k1);
    if (has_next) {  // issue next tile's global loads early
      st0 = bf16x8{};
      st1 = bf16x8{};
      if (s_alive) {
        st0 = *reinterpret_cast<const bf16x8*>(s_xp + k + KC + s_col8 * 8);
        st1 = *reinterpret_cast<const bf16x8*>(s_xp + k + KC + 32 + s_col8 * 8);
      }
    }
#pragma unroll
    for (int kk = 0; kk < KC; kk += 32) {
      const long koff = k + kk + lg * 8;
      // weights stream once per CU: non-temporal (guide: nt-weights)
      bf16v8 bfrag = __builtin_nontemporal_load(
          reinterpret_cast<const bf16v8*>(wp + koff));
      bf16v8 a[4];
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        a[ms] = *reinterpret_cast<const bf16v8*>(
            &x_lds[buf][(ms * 16 + l16) * XS + kk + lg * 8]);
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        acc[ms] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[ms], bfrag,
                                                          acc[ms], 0, 0, 0);
    }
    if (has_next) {  // write pass after the MFMAs (T14)
      *reinterpret_cast<bf16x8*>(&x_lds[buf ^ 1][s_row * XS + s_col8 * 8]) = st0;
      *reinterpret_cast<bf16x8*>(&x_lds[buf ^ 1][s_row * XS + 32 + s_col8 * 8]) = st1;
    }
    __syncthreads();
    buf ^= 1;
  }

  // epilogue: C lane holds rows (lg*4 + r) of each m-sub, col l16
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = ms * 16 + lg * 4 + r;
      if (row >= M) continue;
      const int col = blockIdx.x * 64 + (threadIdx.x / WAVE) * 16 + l16;
      if (SPLIT) {
        float* o = reinterpret_cast<float*>(out);
        o[((long)split * M + row) * N + col] = acc[ms][r];
      } else {
        short* o = reinterpret_cast<short*>(out);
        o[(long)row * N + col] = f2bits(acc[ms][r]);
      }
    }
  }
}

__global__ __launch_bounds__(256) void splitk_combine_kernel(
    short* __restrict__ out, const float* __restrict__ ws, long mn, int S) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= mn) return;
  float acc = 0.f;
  for (int s = 0; s < S; ++s) acc += ws[(long)s * mn + i];
  out[i] = f2bits(acc);
}

}  // namespace

void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor ws, long split) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  TORCH_CHECK(M <= 64, "skinny_gemm: M <= 64");
  TORCH_CHECK(N % 64 == 0, "skinny_gemm: N % 64 == 0");
  TORCH_CHECK(K % 64 == 0, "skinny_gemm: K % 64 == 0");
  auto stream = at::hip::getCurrentHIPStream();
  const int ntiles = N / 64;
  TORCH_CHECK(split >= 1 && (K % (64 * split)) == 0,
              "invalid split for K (k slices must be 64-aligned)");
  const int kps = K / (int)split;
  const bool full = (M == 64);
#define SG_LAUNCH(SPLIT_, FULL_, OUTP)                                         \
  hipLaunchKernelGGL((skinny_gemm_kernel<SPLIT_, FULL_>),                      \
                     dim3(ntiles, SPLIT_ ? (int)split : 1), dim3(256), 0,      \
                     stream, OUTP, (const short*)x.data_ptr(),                 \
                     (const short*)w.data_ptr(), M, N, K, kps)
  if (split == 1) {
    if (full) SG_LAUNCH(false, true, out.data_ptr());
    else SG_LAUNCH(false, false, out.data_ptr());
  } else {
    TORCH_CHECK(ws.numel() >= (long)split * M * N,
                "skinny_gemm split-K workspace too small");
    TORCH_CHECK(ws.scalar_type() == at::kFloat);
    if (full) SG_LAUNCH(true, true, ws.data_ptr());
    else SG_LAUNCH(true, false, ws.data_ptr());
    const long mn = (long)M * N;
    hipLaunchKernelGGL(splitk_combine_kernel,
                       dim3((mn + 255) / 256), dim3(256), 0, stream,
                       (short*)out.data_ptr(), ws.data_ptr<float>(), mn, split);
  }
#undef SG_LAUNCH
}
