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

  // x tile [64][KC] staged through LDS once per BLOCK per k-chunk (waves
  // reading x straight from global exceeded the per-CU L2 bandwidth).
  // Double-buffered, T14 split: issue next tile's loads early, ds_write
  // after the MFMAs (guide §6 G15). KC=256 so each wave holds EIGHT
  // in-flight nt W loads per chunk (deep unroll, late vmcnt — the
  // decode-weights idiom): ~8 KB of W in flight per CU covers HBM latency.
  // Row stride 264 shorts: fragment-read bank offsets row*132 dwords
  // (mod 64 = row*4) are distinct over a 16-lane group — conflict-free.
  constexpr int KC = 256;
  constexpr int XS = KC + 8;
  __shared__ short x_lds[2][64 * XS];
  const int s_row = threadIdx.x % 64;        // staging: this thread's x row
  const int s_col0 = (threadIdx.x / 64) * 8; // + 32*i, i in 0..7
  const short* s_xp = x + (long)(FULL ? s_row : min(s_row, M - 1)) * K;
  const bool s_alive = FULL || s_row < M;

  bf16x8 st[8];
  auto stage_load = [&](int k) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      st[i] = bf16x8{};
      if (s_alive)
        st[i] = *reinterpret_cast<const bf16x8*>(s_xp + k + s_col0 + 32 * i);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<bf16x8*>(
          &x_lds[buf][s_row * XS + s_col0 + 32 * i]) = st[i];
  };

  stage_load(k0);
  stage_write(0);
  __syncthreads();

  int buf = 0;
  for (int k = k0; k < k1; k += KC) {
    const bool has_next = (k + KC < k1);
    if (has_next) stage_load(k + KC);  // x loads for tile t+1 in flight
    // all 8 W fragments of this chunk issued before any MFMA consumes one
    bf16v8 bw[8];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      bw[i] = __builtin_nontemporal_load(
          reinterpret_cast<const bf16v8*>(wp + k + i * 32 + lg * 8));
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      bf16v8 a[4];
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        a[ms] = *reinterpret_cast<const bf16v8*>(
            &x_lds[buf][(ms * 16 + l16) * XS + i * 32 + lg * 8]);
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        acc[ms] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[ms], bw[i],
                                                          acc[ms], 0, 0, 0);
    }
    if (has_next) stage_write(buf ^ 1);  // write pass after the MFMAs (T14)
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
  // 4 elements per thread, 16-B loads per plane: the scalar version ran
  // latency-bound at ~0.8 TB/s and cost 4.8 us inside the decode graph
  const long i4 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i4 >= mn) return;
  if (i4 + 4 <= mn) {
    f32x4 acc = *reinterpret_cast<const f32x4*>(ws + i4);
    for (int s = 1; s < S; ++s) {
      const f32x4 v = *reinterpret_cast<const f32x4*>(ws + (long)s * mn + i4);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += v[j];
    }
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bits(acc[j]);
    *reinterpret_cast<bf16x4*>(out + i4) = o;
  } else {
    for (long i = i4; i < mn; ++i) {
      float acc = 0.f;
      for (int s = 0; s < S; ++s) acc += ws[(long)s * mn + i];
      out[i] = f2bits(acc);
    }
  }
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
  TORCH_CHECK(K % 256 == 0, "skinny_gemm: K % 256 == 0");
  auto stream = at::hip::getCurrentHIPStream();
  const int ntiles = N / 64;
  TORCH_CHECK(split >= 1 && (K % (256 * split)) == 0,
              "invalid split for K (k slices must be 256-aligned)");
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
                       dim3((mn + 1023) / 1024), dim3(256), 0, stream,
                       (short*)out.data_ptr(), ws.data_ptr<float>(), mn, split);
  }
#undef SG_LAUNCH
}

// ---------------------------------------------------------------------------
// Packed-weight variant: W pre-shuffled at model load into fragment-linear
// layout packed[n_tile][k_chunk][lane][8] (ops.pack_weight), so each wave
// streams its 16-row x K weight block CONTIGUOUSLY — 1 KB per load
// instruction, one stream per wave. This is what buys the HBM roofline;
// the unpacked variant's 16 scattered 64-B row segments per instruction
// cap out around 2-3 TB/s.

namespace {

template <bool SPLIT, bool FULL, bool NT, int KC>
__global__ __launch_bounds__(256) void skinny_gemm_packed_kernel(
    void* __restrict__ out, const short* __restrict__ x,
    const short* __restrict__ wp_packed, int M, int N, int K,
    int k_per_split) {
  const int n_tile = blockIdx.x * 4 + threadIdx.x / WAVE;  // 16-col tile
  const int split = blockIdx.y;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane % 16;
  const int lg = lane / 16;
  const int k0 = split * k_per_split;
  const int k1 = min(k0 + k_per_split, K);

  f32x4 acc[4];
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) acc[ms] = f32x4{0.f, 0.f, 0.f, 0.f};

  // this wave's contiguous weight stream
  const short* wp = wp_packed + ((long)n_tile * (K / 32) + k0 / 32) * 512 +
                    lane * 8;

  // KC=256: deep per-wave pipeline (8 W frags in flight) at 67.6 KB LDS
  // (1-2 blocks/CU). KC=128: half the LDS and W registers so 2x the
  // blocks/CU carry the HBM latency instead of per-wave depth.
  constexpr int FR = KC / 32;  // W fragments / x vectors per chunk
  constexpr int XS = KC + 8;
  __shared__ short x_lds[2][64 * XS];
  const int s_row = threadIdx.x % 64;
  const int s_col0 = (threadIdx.x / 64) * 8;
  const short* s_xp = x + (long)(FULL ? s_row : min(s_row, M - 1)) * K;
  const bool s_alive = FULL || s_row < M;

  bf16x8 st[FR];
  auto stage_load = [&](int k) {
#pragma unroll
    for (int i = 0; i < FR; ++i) {
      st[i] = bf16x8{};
      if (s_alive)
        st[i] = *reinterpret_cast<const bf16x8*>(s_xp + k + s_col0 + 32 * i);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int i = 0; i < FR; ++i)
      *reinterpret_cast<bf16x8*>(
          &x_lds[buf][s_row * XS + s_col0 + 32 * i]) = st[i];
  };

  // W stream is software-pipelined ACROSS the per-chunk barrier: chunk
  // t+1's eight nt loads are issued while chunk t computes, so no chunk
  // pays the ~900-cycle HBM latency cold (the barrier would otherwise
  // serialize it at 1-2 blocks/CU).
  bf16v8 bw_cur[FR], bw_nxt[FR];
  auto w_load = [&](bf16v8 (&dst)[FR], long woff) {
#pragma unroll
    for (int i = 0; i < FR; ++i) {
      const bf16v8* p = reinterpret_cast<const bf16v8*>(wp + woff + (long)i * 512);
      dst[i] = NT ? __builtin_nontemporal_load(p) : *p;
    }
  };

  auto compute = [&](bf16v8 (&bw)[FR], int buf) {
#pragma unroll
    for (int i = 0; i < FR; ++i) {
      bf16v8 a[4];
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        a[ms] = *reinterpret_cast<const bf16v8*>(
            &x_lds[buf][(ms * 16 + l16) * XS + i * 32 + lg * 8]);
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        acc[ms] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[ms], bw[i],
                                                          acc[ms], 0, 0, 0);
    }
  };
  stage_load(k0);
  stage_write(0);
  w_load(bw_cur, 0);
  __syncthreads();

  // ping-pong 2x unroll: no bw register copies, so chunk t+1's weight
  // loads never drain at a chunk boundary (a copy would force vmcnt(0))
  int buf = 0;
  long woff = (long)FR * 512;
  int k = k0;
  while (true) {
    bool has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_nxt, woff);
      woff += (long)FR * 512;
    }
    compute(bw_cur, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
    has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_cur, woff);
      woff += (long)FR * 512;
    }
    compute(bw_nxt, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
  }

#pragma unroll
  for (int ms = 0; ms < 4; ++ms) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = ms * 16 + lg * 4 + r;
      if (row >= M) continue;
      const int col = n_tile * 16 + l16;
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

}  // namespace

void skinny_gemm_packed(torch::Tensor out, torch::Tensor x,
                        torch::Tensor w_packed, long N, long K,
                        torch::Tensor ws, long split, bool nt, long kc) {
  TORCH_CHECK(x.is_contiguous() && w_packed.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w_packed.scalar_type() == at::kBFloat16);
  const int M = x.size(0);
  TORCH_CHECK(x.size(1) == K);
  TORCH_CHECK(w_packed.numel() == N * K, "packed weight numel mismatch");
  TORCH_CHECK(M <= 64 && N % 64 == 0 && K % 256 == 0);
  TORCH_CHECK(split >= 1 && (K % (256 * split)) == 0);
  auto stream = at::hip::getCurrentHIPStream();
  const int ntiles = (int)N / 64;
  const int kps = (int)K / (int)split;
  const bool full = (M == 64);
  TORCH_CHECK(kc == 128 || kc == 256, "kc must be 128 or 256");
  TORCH_CHECK((K % (kc * split)) == 0, "K must be kc*split aligned");
#define SGP_LAUNCH(SPLIT_, FULL_, NT_, KC_, OUTP)                              \
  hipLaunchKernelGGL((skinny_gemm_packed_kernel<SPLIT_, FULL_, NT_, KC_>),     \
                     dim3(ntiles, SPLIT_ ? (int)split : 1), dim3(256), 0,      \
                     stream, OUTP, (const short*)x.data_ptr(),                 \
                     (const short*)w_packed.data_ptr(), M, (int)N, (int)K, kps)
#define SGP_KC(SPLIT_, FULL_, NT_, OUTP)                                       \
  do {                                                                         \
    if (kc == 128) SGP_LAUNCH(SPLIT_, FULL_, NT_, 128, OUTP);                  \
    else SGP_LAUNCH(SPLIT_, FULL_, NT_, 256, OUTP);                            \
  } while (0)
#define SGP_DISPATCH(SPLIT_, OUTP)                                             \
  do {                                                                         \
    if (full && nt) SGP_KC(SPLIT_, true, true, OUTP);                          \
    else if (full) SGP_KC(SPLIT_, true, false, OUTP);                          \
    else if (nt) SGP_KC(SPLIT_, false, true, OUTP);                            \
    else SGP_KC(SPLIT_, false, false, OUTP);                                   \
  } while (0)
  if (split == 1) {
    SGP_DISPATCH(false, out.data_ptr());
  } else {
    TORCH_CHECK(ws.numel() >= (long)split * M * N && ws.scalar_type() == at::kFloat);
    SGP_DISPATCH(true, ws.data_ptr());
    const long mn = (long)M * N;
    hipLaunchKernelGGL(splitk_combine_kernel, dim3((mn + 1023) / 1024),
                       dim3(256), 0, stream, (short*)out.data_ptr(),
                       ws.data_ptr<float>(), mn, split);
  }
#undef SGP_DISPATCH
#undef SGP_LAUNCH
}

// ---------------------------------------------------------------------------
// fp8 (OCP e4m3) skinny GEMM — the Mixtral expert decode GEMM of
// BASELINE.json config 5. Same structure as the bf16 packed kernel, but
// operands are 8 fp8 bytes per lane (v_mfma_f32_16x16x32_fp8_fp8, i64
// fragments) so the expert weight stream is HALF the bytes of bf16.
// Scales: per-output-channel weight scales sw[n] (offline), per-token
// activation scales sx[m] (quant_fp8_rows, dynamic) — applied in the
// epilogue: out = acc * sx[row] * sw[col].

#include <hip/hip_fp8.h>

namespace {

// per-row dynamic quantization: x bf16 [M,K] -> x8 e4m3 [M,K] + sx f32 [M]
__global__ __launch_bounds__(256) void quant_fp8_rows_kernel(
    unsigned char* __restrict__ x8, float* __restrict__ sx,
    const short* __restrict__ x, int K) {
  const int row = blockIdx.x;
  const short* xp = x + (long)row * K;
  float mx = 0.f;
  for (int i = threadIdx.x * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) mx = fmaxf(mx, fabsf(bits2f(v[j])));
  }
  mx = wave_max(mx);
  __shared__ float red[4];
  if ((threadIdx.x & 63) == 0) red[threadIdx.x / 64] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  const float scale = fmaxf(mx, 1e-8f) / 448.f;  // e4m3 max normal = 448
  const float inv = 1.0f / scale;
  if (threadIdx.x == 0) sx[row] = scale;
  for (int i = threadIdx.x * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xp + i);
    unsigned char o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 f8(bits2f(v[j]) * inv);
      o[j] = f8.__x;
    }
    *reinterpret_cast<uint2*>(x8 + (long)row * K + i) =
        *reinterpret_cast<uint2*>(o);
  }
}

template <bool SPLIT>
__global__ __launch_bounds__(256) void skinny_gemm_fp8_kernel(
    void* __restrict__ out,                 // bf16 [M,N] or f32 ws
    const unsigned char* __restrict__ x8,   // [M,K] e4m3
    const float* __restrict__ sx,           // [M]
    const unsigned char* __restrict__ wp_packed,  // fragment-linear e4m3
    const float* __restrict__ sw,           // [N]
    int M, int N, int K, int k_per_split) {
  const int n_tile = blockIdx.x * 4 + threadIdx.x / WAVE;
  const int split = blockIdx.y;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane % 16;
  const int lg = lane / 16;
  const int k0 = split * k_per_split;
  const int k1 = min(k0 + k_per_split, K);

  f32x4 acc[4];
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) acc[ms] = f32x4{0.f, 0.f, 0.f, 0.f};

  const unsigned char* wp =
      wp_packed + ((long)n_tile * (K / 32) + k0 / 32) * 512 + lane * 8;

  constexpr int KC = 256;
  constexpr int XS = KC + 16;  // bytes; keeps 8-B frag reads conflict-light
  __shared__ unsigned char x_lds[2][64 * XS];
  const int s_row = threadIdx.x % 64;
  const int s_col0 = (threadIdx.x / 64) * 8;  // + 32*i, i in 0..7 (bytes)
  const unsigned char* s_xp = x8 + (long)min(s_row, M - 1) * K;
  const bool s_alive = s_row < M;

  uint2 st[8];
  auto stage_load = [&](int k) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      st[i] = uint2{0, 0};
      if (s_alive)
        st[i] = *reinterpret_cast<const uint2*>(s_xp + k + s_col0 + 32 * i);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<uint2*>(&x_lds[buf][s_row * XS + s_col0 + 32 * i]) = st[i];
  };

  long bw_cur[8], bw_nxt[8];
  auto w_load = [&](long (&dst)[8], long woff) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      dst[i] = __builtin_nontemporal_load(
          reinterpret_cast<const long*>(wp + woff + (long)i * 512));
  };

  auto compute = [&](long (&bw)[8], int buf) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      long a[4];
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        a[ms] = *reinterpret_cast<const long*>(
            &x_lds[buf][(ms * 16 + l16) * XS + i * 32 + lg * 8]);
#pragma unroll
      for (int ms = 0; ms < 4; ++ms)
        acc[ms] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            a[ms], bw[i], acc[ms], 0, 0, 0);
    }
  };
  stage_load(k0);
  stage_write(0);
  w_load(bw_cur, 0);
  __syncthreads();

  int buf = 0;
  long woff = 8 * 512;
  int k = k0;
  while (true) {
    bool has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_nxt, woff);
      woff += 8 * 512;
    }
    compute(bw_cur, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
    has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_cur, woff);
      woff += 8 * 512;
    }
    compute(bw_nxt, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
  }

  const int col = n_tile * 16 + l16;
  const float swc = sw[col];
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = ms * 16 + lg * 4 + r;
      if (row >= M) continue;
      const float v = acc[ms][r] * sx[row] * swc;
      if (SPLIT) {
        reinterpret_cast<float*>(out)[((long)split * M + row) * N + col] = v;
      } else {
        reinterpret_cast<short*>(out)[(long)row * N + col] = f2bits(v);
      }
    }
  }
}

}  // namespace


namespace {

// e2m1 encode: nearest of {0,.5,1,1.5,2,3,4,6} with sign bit 3
__device__ __forceinline__ unsigned char f2e2m1(float v) {
  const float a = fabsf(v);
  unsigned char c;
  if (a < 0.25f) c = 0;
  else if (a < 0.75f) c = 1;
  else if (a < 1.25f) c = 2;
  else if (a < 1.75f) c = 3;
  else if (a < 2.5f) c = 4;
  else if (a < 3.5f) c = 5;
  else if (a < 5.0f) c = 6;
  else c = 7;
  return c | (v < 0.f ? 8 : 0);
}

// per-row dynamic fp4 quantization: x bf16 [M,K] -> x4 [M,K/2] (2 codes
// per byte, lo nibble = even k) + sx f32 [M] with amax -> 6.0
__global__ __launch_bounds__(256) void quant_fp4_rows_kernel(
    unsigned char* __restrict__ x4, float* __restrict__ sx,
    const short* __restrict__ x, int K) {
  const int row = blockIdx.x;
  const short* xp = x + (long)row * K;
  float mx = 0.f;
  for (int i = threadIdx.x * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) mx = fmaxf(mx, fabsf(bits2f(v[j])));
  }
  mx = wave_max(mx);
  __shared__ float red[4];
  if ((threadIdx.x & 63) == 0) red[threadIdx.x / 64] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  const float scale = fmaxf(mx, 1e-8f) / 6.f;  // e2m1 max = 6
  const float inv = 1.0f / scale;
  if (threadIdx.x == 0) sx[row] = scale;
  for (int i = threadIdx.x * 8; i < K; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xp + i);
    unsigned char o[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      o[j] = f2e2m1(bits2f(v[2 * j]) * inv) |
             (f2e2m1(bits2f(v[2 * j + 1]) * inv) << 4);
    *reinterpret_cast<unsigned int*>(x4 + ((long)row * K + i) / 2) =
        *reinterpret_cast<unsigned int*>(o);
  }
}

}  // namespace

void quant_fp4_rows(torch::Tensor x4, torch::Tensor sx, torch::Tensor x) {
  TORCH_CHECK(x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x4.scalar_type() == at::kByte && sx.scalar_type() == at::kFloat);
  const int M = x.size(0), K = x.size(1);
  TORCH_CHECK(K % 16 == 0 && x4.numel() == (long)M * K / 2);
  auto stream = at::hip::getCurrentHIPStream();
  if (M == 0) return;
  hipLaunchKernelGGL(quant_fp4_rows_kernel, dim3(M), dim3(256), 0, stream,
                     x4.data_ptr<uint8_t>(), sx.data_ptr<float>(),
                     (const short*)x.data_ptr(), K);
}

// ---------------------------------------------------------------------------
// MXFP4 expert GEMM (gfx950 block-scaled MFMA): W stored as e2m1 codes
// (2/byte) with one e8m0 scale per (col, 64-k block); activations e4m3
// per-row (quant_fp8_rows). One v_mfma_scale_f32_16x16x128_f8f6f4 per
// 128-k slice — QUARTER the weight bytes of bf16. Operand lane maps were
// established empirically (tools/mx_probe.py):
//   A (fp8): lane l holds A[l%16][(l/16)*32 .. +32]  (32 bytes)
//   B (fp4): lane l holds B[(l/16)*32 + j][l%16], 32 codes = 16 bytes in
//            the LOW half of the 8xi32 operand
//   scales:  byte 0 of the scale int; lanes 0..31 cover (row|col = l%16,
//            64-elem k-half = l/16); lanes 32..63 mirror (fed 127 = x1)
//   C:       lane l holds rows (l/16)*4 + r, col l%16

namespace {

typedef int i32x8v __attribute__((ext_vector_type(8)));
typedef int i32x4v __attribute__((ext_vector_type(4)));

template <bool SPLIT, int COMBO>
__global__ __launch_bounds__(256) void skinny_gemm_mxfp4_kernel(
    void* __restrict__ out, const unsigned char* __restrict__ x4,
    const float* __restrict__ sx, const unsigned char* __restrict__ wp_packed,
    const unsigned char* __restrict__ wsc,  // [T][K/128][4][16] e8m0
    int M, int N, int K, int k_per_split) {
  const int n_tile = blockIdx.x * 4 + threadIdx.x / WAVE;
  const int split = blockIdx.y;
  const int lane = threadIdx.x % WAVE;
  const int l16 = lane % 16;
  const int lg = lane / 16;
  const int k0 = split * k_per_split;
  const int k1 = min(k0 + k_per_split, K);
  const int KC128 = K / 128;

  f32x4 acc[4];
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) acc[ms] = f32x4{0.f, 0.f, 0.f, 0.f};

  // per-(tile, k128) fragment: lane-major 16 B codes
  const unsigned char* wp =
      wp_packed + (((long)n_tile * KC128 + k0 / 128) * 64 + lane) * 16;
  const unsigned char* scp = wsc + ((long)n_tile * KC128 + k0 / 128) * 64;

  constexpr int KC = 256;              // k per chunk (codes: KC/2 bytes)
  constexpr int XS = KC / 2 + 8;       // bytes per LDS row
  __shared__ unsigned char x_lds[2][64 * XS];
  const int s_row = threadIdx.x % 64;
  const int s_col0 = (threadIdx.x / 64) * 4;   // byte cols, stride 16
  const unsigned char* s_xp = x4 + (long)min(s_row, M - 1) * (K / 2);
  const bool s_alive = s_row < M;

  unsigned int st[8];
  auto stage_load = [&](int k) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      st[i] = 0;
      if (s_alive)
        st[i] = *reinterpret_cast<const unsigned int*>(
            s_xp + k / 2 + s_col0 + 16 * i);
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<unsigned int*>(
          &x_lds[buf][s_row * XS + s_col0 + 16 * i]) = st[i];
  };

  i32x4v bw_cur[2], bw_nxt[2];
  int sc_cur[2], sc_nxt[2];
  auto w_load = [&](i32x4v (&dst)[2], int (&sc)[2], long chunk) {
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      dst[f] = __builtin_nontemporal_load(reinterpret_cast<const i32x4v*>(
          wp + (chunk * 2 + f) * 1024));
      // fp4 scale-lane map is bijective: lane l = (kb l/16, col l%16)
      sc[f] = (int)scp[(chunk * 2 + f) * 64 + lane];
    }
  };

  // the scale operands must BOTH be VGPRs: with a literal/SGPR in either
  // slot the compiler selects an encoding whose scales are inert
  // (probe-verified; tools/mx_diag.py)
  int unit_sc;
  asm volatile("v_mov_b32 %0, 127" : "=v"(unit_sc));
  auto compute = [&](i32x4v (&bw)[2], int (&sc)[2], int buf) {
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      i32x8v b8 = {bw[f][0], bw[f][1], bw[f][2], bw[f][3], 0, 0, 0, 0};
#pragma unroll
      for (int ms = 0; ms < 4; ++ms) {
        // A fp4: lane's 32 codes = 16 bytes, LOW half of the operand
        const i32x4v a4 = *reinterpret_cast<const i32x4v*>(
            &x_lds[buf][(ms * 16 + l16) * XS + f * 64 + lg * 16]);
        i32x8v a8 = {a4[0], a4[1], a4[2], a4[3], 0, 0, 0, 0};
        if constexpr (COMBO == 1)
          acc[ms] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
              a8, b8, acc[ms], 4, 4, 0, sc[f], 0, unit_sc);
        else
          acc[ms] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
              a8, b8, acc[ms], 4, 4, 0, unit_sc, 0, sc[f]);
      }
    }
  };

  stage_load(k0);
  stage_write(0);
  w_load(bw_cur, sc_cur, 0);
  __syncthreads();

  int buf = 0;
  long chunk = 1;
  int k = k0;
  while (true) {
    bool has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_nxt, sc_nxt, chunk++);
    }
    compute(bw_cur, sc_cur, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
    has_next = (k + KC < k1);
    if (has_next) {
      stage_load(k + KC);
      w_load(bw_cur, sc_cur, chunk++);
    }
    compute(bw_nxt, sc_nxt, buf);
    if (has_next) stage_write(buf ^ 1);
    __syncthreads();
    buf ^= 1;
    k += KC;
    if (!has_next) break;
  }

  const int col = n_tile * 16 + l16;
#pragma unroll
  for (int ms = 0; ms < 4; ++ms) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = ms * 16 + lg * 4 + r;
      if (row >= M) continue;
      const float v = acc[ms][r] * sx[row];
      if (SPLIT) {
        reinterpret_cast<float*>(out)[((long)split * M + row) * N + col] = v;
      } else {
        reinterpret_cast<short*>(out)[(long)row * N + col] = f2bits(v);
      }
    }
  }
}

}  // namespace

void skinny_gemm_mxfp4(torch::Tensor out, torch::Tensor x8, torch::Tensor sx,
                       torch::Tensor w_packed, torch::Tensor w_scales, long N,
                       long K, torch::Tensor ws, long split, long combo) {
  TORCH_CHECK(x8.is_contiguous() && w_packed.is_contiguous());
  TORCH_CHECK(x8.scalar_type() == at::kByte &&
              w_packed.scalar_type() == at::kByte &&
              w_scales.scalar_type() == at::kByte);
  const int M = x8.size(0);
  TORCH_CHECK(x8.size(1) == K / 2 && K % 256 == 0 && N % 64 == 0 && M <= 64);
  TORCH_CHECK(w_packed.numel() == N * K / 2, "mxfp4 pack numel");
  TORCH_CHECK(w_scales.numel() == (N / 16) * (K / 128) * 64, "mxfp4 scales");
  TORCH_CHECK(split >= 1 && (K % (256 * split)) == 0);
  auto stream = at::hip::getCurrentHIPStream();
  const int ntiles = (int)N / 64;
  const int kps = (int)K / (int)split;
#define MX_LAUNCH(SPLIT_, COMBO_, OUTP)                                        \
  hipLaunchKernelGGL((skinny_gemm_mxfp4_kernel<SPLIT_, COMBO_>),               \
                     dim3(ntiles, SPLIT_ ? (int)split : 1), dim3(256), 0,      \
                     stream, OUTP, x8.data_ptr<uint8_t>(),                     \
                     sx.data_ptr<float>(), w_packed.data_ptr<uint8_t>(),       \
                     w_scales.data_ptr<uint8_t>(), M, (int)N, (int)K, kps)
#define MX_COMBO(SPLIT_, OUTP)                                                 \
  do {                                                                         \
    if (combo == 1) MX_LAUNCH(SPLIT_, 1, OUTP);                                \
    else MX_LAUNCH(SPLIT_, 0, OUTP);                                          \
  } while (0)
  if (split == 1) {
    MX_COMBO(false, out.data_ptr());
  } else {
    TORCH_CHECK(ws.numel() >= (long)split * M * N &&
                ws.scalar_type() == at::kFloat);
    MX_COMBO(true, ws.data_ptr());
    const long mn = (long)M * N;
    hipLaunchKernelGGL(splitk_combine_kernel, dim3((mn + 1023) / 1024),
                       dim3(256), 0, stream, (short*)out.data_ptr(),
                       ws.data_ptr<float>(), mn, split);
  }
#undef MX_COMBO
#undef MX_LAUNCH
}

void quant_fp8_rows(torch::Tensor x8, torch::Tensor sx, torch::Tensor x) {
  TORCH_CHECK(x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x8.scalar_type() == at::kByte && sx.scalar_type() == at::kFloat);
  const int M = x.size(0), K = x.size(1);
  TORCH_CHECK(K % 8 == 0);
  if (M == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(quant_fp8_rows_kernel, dim3(M), dim3(256), 0, stream,
                     x8.data_ptr<unsigned char>(), sx.data_ptr<float>(),
                     (const short*)x.data_ptr(), K);
}

void skinny_gemm_fp8(torch::Tensor out, torch::Tensor x8, torch::Tensor sx,
                     torch::Tensor w_packed, torch::Tensor sw, long N, long K,
                     torch::Tensor ws, long split) {
  TORCH_CHECK(x8.is_contiguous() && w_packed.is_contiguous());
  TORCH_CHECK(x8.scalar_type() == at::kByte &&
              w_packed.scalar_type() == at::kByte);
  TORCH_CHECK(sw.scalar_type() == at::kFloat && sx.scalar_type() == at::kFloat);
  const int M = x8.size(0);
  TORCH_CHECK(x8.size(1) == K);
  TORCH_CHECK(M <= 64 && N % 64 == 0 && K % 256 == 0);
  TORCH_CHECK(split >= 1 && (K % (256 * split)) == 0);
  auto stream = at::hip::getCurrentHIPStream();
  const int ntiles = (int)N / 64;
  const int kps = (int)K / (int)split;
  if (split == 1) {
    hipLaunchKernelGGL((skinny_gemm_fp8_kernel<false>), dim3(ntiles, 1),
                       dim3(256), 0, stream, out.data_ptr(),
                       x8.data_ptr<unsigned char>(), sx.data_ptr<float>(),
                       w_packed.data_ptr<unsigned char>(), sw.data_ptr<float>(),
                       M, (int)N, (int)K, kps);
  } else {
    TORCH_CHECK(ws.numel() >= (long)split * M * N &&
                ws.scalar_type() == at::kFloat);
    hipLaunchKernelGGL((skinny_gemm_fp8_kernel<true>), dim3(ntiles, (int)split),
                       dim3(256), 0, stream, ws.data_ptr(),
                       x8.data_ptr<unsigned char>(), sx.data_ptr<float>(),
                       w_packed.data_ptr<unsigned char>(), sw.data_ptr<float>(),
                       M, (int)N, (int)K, kps);
    const long mn = (long)M * N;
    hipLaunchKernelGGL(splitk_combine_kernel, dim3((mn + 1023) / 1024),
                       dim3(256), 0, stream, (short*)out.data_ptr(),
                       ws.data_ptr<float>(), mn, split);
  }
}
