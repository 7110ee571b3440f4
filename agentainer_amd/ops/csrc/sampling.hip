// Sampling kernels (gfx950): greedy argmax + temperature/top-p categorical.
//
// Serves /chat response generation (SURVEY.md §2.3 "Sampling kernel" row;
// greedy is required for BASELINE config-2 parity). Vocab ~128k bf16
// logits per row — both kernels stream the row once per pass with 16-B
// vectorized loads.
//
// topp_sample: per row — temperature scale, streaming max + exp-sum
// (softmax denominator), then ADAPTIVE candidate harvest: token counts are
// taken at four probability floors (p >= {1e-1,1e-2,1e-3,1e-4} * p_max) in
// one pass and the loosest floor whose candidate set fits the LDS buffer
// is harvested, bitonic-sorted by probability, nucleus-cut at top_p and
// drawn by inverse CDF. Exact whenever the nucleus lies above the chosen
// floor (always, for peaked LLM logits); for pathologically flat rows the
// draw truncates to the top-CAND_CAP tokens (a p<=1e-1*p_max tail), and if
// even the tightest floor overflows it falls back to a full categorical
// draw without the nucleus cut.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cfloat>
#include "common.h"

namespace {

constexpr int CAND_CAP = 512;
constexpr float P_FLOOR = 1e-4f;  // harvest tokens with p >= P_FLOOR * max

__global__ __launch_bounds__(256) void greedy_kernel(
    long* __restrict__ out, const short* __restrict__ logits, int V) {
  const int row = blockIdx.x;
  const short* lp = logits + (long)row * V;
  float best = -FLT_MAX;
  int best_i = 0;
  const int tid = threadIdx.x;
  for (int i = tid * 8; i + 7 < V; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(lp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bits2f(v[j]);
      if (f > best || (f == best && i + j < best_i)) {
        best = f;
        best_i = i + j;
      }
    }
  }
  for (int i = (V / 8) * 8 + tid; i < V; i += blockDim.x) {
    float f = bits2f(lp[i]);
    if (f > best) { best = f; best_i = i; }
  }
  // wave reduce (value, index) — ties resolve to the smallest index
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, WAVE);
    int oi = __shfl_xor(best_i, off, WAVE);
    if (ov > best || (ov == best && oi < best_i)) { best = ov; best_i = oi; }
  }
  __shared__ float sv[4];
  __shared__ int si[4];
  if ((tid & 63) == 0) { sv[tid / 64] = best; si[tid / 64] = best_i; }
  __syncthreads();
  if (tid == 0) {
    best = sv[0]; best_i = si[0];
#pragma unroll
    for (int w = 1; w < 4; ++w)
      if (sv[w] > best || (sv[w] == best && si[w] < best_i)) {
        best = sv[w]; best_i = si[w];
      }
    out[row] = best_i;
  }
}

// xorshift128 per row seeded from (seed, row) — deterministic across replays
__device__ __forceinline__ float uniform01(unsigned long long seed, int row) {
  unsigned long long x = seed ^ (0x9e3779b97f4a7c15ull * (row + 1));
  x ^= x >> 12; x ^= x << 25; x ^= x >> 27;
  x *= 0x2545F4914F6CDD1Dull;
  return (float)((x >> 40) & 0xFFFFFF) / 16777216.0f;
}

__global__ __launch_bounds__(256) void topp_kernel(
    long* __restrict__ out, const short* __restrict__ logits,
    const float* __restrict__ temps, const float* __restrict__ top_ps,
    const unsigned long long* __restrict__ seeds, int V) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const short* lp = logits + (long)row * V;
  const float invT = 1.0f / fmaxf(temps[row], 1e-6f);
  const float top_p = top_ps[row];

  __shared__ float red[4];
  __shared__ float cand_p[CAND_CAP];
  __shared__ int cand_i[CAND_CAP];
  __shared__ int n_cand;
  __shared__ int tier_counts[4];
  __shared__ float chosen_floor_s;
  __shared__ float row_max_s, row_sum_s;

  // pass 1: max of scaled logits
  float mx = -FLT_MAX;
  for (int i = tid * 8; i + 7 < V; i += blockDim.x * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(lp + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) mx = fmaxf(mx, bits2f(v[j]) * invT);
  }
  for (int i = (V / 8) * 8 + tid; i < V; i += blockDim.x)
    mx = fmaxf(mx, bits2f(lp[i]) * invT);
  mx = wave_max(mx);
  if ((tid & 63) == 0) red[tid / 64] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  if (tid == 0) {
    row_max_s = mx;
    n_cand = 0;
    for (int t = 0; t < 4; ++t) tier_counts[t] = 0;
  }
  __syncthreads();

  // pass 2: exp-sum + candidate counts at the four floors
  // floors: p >= 10^-(t+1) * p_max  =>  s - mx >= -(t+1)*ln(10)
  const float LN10 = 2.302585093f;
  float sum = 0.f;
  int my_counts[4] = {0, 0, 0, 0};
  for (int i = tid; i < V; i += blockDim.x) {
    float s = bits2f(lp[i]) * invT - mx;
    sum += __expf(s);
#pragma unroll
    for (int t = 0; t < 4; ++t)
      if (s >= -(float)(t + 1) * LN10) ++my_counts[t];
  }
  sum = wave_sum(sum);
  if ((tid & 63) == 0) red[tid / 64] = sum;
#pragma unroll
  for (int t = 0; t < 4; ++t)
    if (my_counts[t]) atomicAdd(&tier_counts[t], my_counts[t]);
  __syncthreads();
  sum = red[0] + red[1] + red[2] + red[3];
  if (tid == 0) {
    row_sum_s = sum;
    // loosest floor (largest t) whose candidate set fits
    float floor_logit = 1.0f;  // sentinel: none fits
    for (int t = 3; t >= 0; --t)
      if (tier_counts[t] <= CAND_CAP) { floor_logit = -(float)(t + 1) * LN10; break; }
    chosen_floor_s = floor_logit;
  }
  __syncthreads();
  const float total = row_sum_s;
  const float u = uniform01(seeds[row], row);
  const float floor_logit = chosen_floor_s;

  if (floor_logit <= 0.f) {
    // pass 3: harvest at the chosen floor
    for (int i = tid; i < V; i += blockDim.x) {
      float s = bits2f(lp[i]) * invT - mx;
      if (s >= floor_logit) {
        int slot = atomicAdd(&n_cand, 1);
        if (slot < CAND_CAP) { cand_p[slot] = __expf(s); cand_i[slot] = i; }
      }
    }
    __syncthreads();
  }

  if (floor_logit <= 0.f && n_cand <= CAND_CAP) {
    // bitonic sort candidates by p descending (padded to pow2 with -1)
    int n = n_cand;
    int npow = 1;
    while (npow < n) npow <<= 1;
    for (int i = tid + n; i < npow; i += blockDim.x) {
      if (i < CAND_CAP) { cand_p[i] = -1.f; cand_i[i] = V; }
    }
    __syncthreads();
    if (npow <= CAND_CAP) {
      for (int k = 2; k <= npow; k <<= 1) {
        for (int j = k >> 1; j > 0; j >>= 1) {
          for (int i = tid; i < npow; i += blockDim.x) {
            int ixj = i ^ j;
            if (ixj > i) {
              bool up = (i & k) == 0;  // descending overall
              float pi = cand_p[i], pj = cand_p[ixj];
              if (up ? (pi < pj) : (pi > pj)) {
                int ti = cand_i[i];
                cand_p[i] = pj; cand_p[ixj] = pi;
                cand_i[i] = cand_i[ixj]; cand_i[ixj] = ti;
              }
            }
          }
          __syncthreads();
        }
      }
      // nucleus cut + inverse-CDF draw (thread 0; n is small)
      if (tid == 0) {
        float mass = 0.f;
        int cut = n;
        for (int i = 0; i < n; ++i) {
          mass += cand_p[i];
          if (mass / total >= top_p) { cut = i + 1; break; }
        }
        float nucleus = 0.f;
        for (int i = 0; i < cut; ++i) nucleus += cand_p[i];
        float target = u * nucleus, acc = 0.f;
        long pick = cand_i[0];
        for (int i = 0; i < cut; ++i) {
          acc += cand_p[i];
          if (acc >= target) { pick = cand_i[i]; break; }
        }
        out[row] = pick;
      }
      return;
    }
  }
  // overflow fallback: full categorical draw without the nucleus cut
  __shared__ float cum_base[4];
  float target = u * total;
  // serial-ish scan: each wave accumulates its stripe; cheap vs correctness
  if (tid == 0) {
    float acc = 0.f;
    long pick = 0;
    for (int i = 0; i < V; ++i) {
      acc += __expf(bits2f(lp[i]) * invT - row_max_s);
      if (acc >= target) { pick = i; break; }
      pick = i;
    }
    out[row] = pick;
  }
  (void)cum_base;
}

}  // namespace

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  TORCH_CHECK(logits.is_contiguous() && logits.scalar_type() == at::kBFloat16);
  TORCH_CHECK(out.scalar_type() == at::kLong);
  const int B = logits.size(0), V = logits.size(1);
  if (B == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(greedy_kernel, dim3(B), dim3(256), 0, stream,
                     out.data_ptr<long>(), (const short*)logits.data_ptr(), V);
}

void topp_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temps,
                 torch::Tensor top_ps, torch::Tensor seeds) {
  TORCH_CHECK(logits.is_contiguous() && logits.scalar_type() == at::kBFloat16);
  TORCH_CHECK(out.scalar_type() == at::kLong);
  TORCH_CHECK(temps.scalar_type() == at::kFloat && top_ps.scalar_type() == at::kFloat);
  const int B = logits.size(0), V = logits.size(1);
  if (B == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(topp_kernel, dim3(B), dim3(256), 0, stream,
                     out.data_ptr<long>(), (const short*)logits.data_ptr(),
                     temps.data_ptr<float>(), top_ps.data_ptr<float>(),
                     (const unsigned long long*)seeds.data_ptr(), V);
}

// ---------------------------------------------------------------------------
// MFMA layout probe (test-only): one v_mfma_f32_16x16x32_bf16 on prepacked
// per-lane fragments. Host packs candidate layouts and checks which
// reconstruction matches a torch matmul — pins down the gfx950 fragment
// maps empirically (used by tools/mfma_probe.py and the kernel tests).
namespace {
typedef __bf16 probe_bf16v8 __attribute__((ext_vector_type(8)));
__global__ void mfma_probe_kernel(const short* __restrict__ a,
                                  const short* __restrict__ b,
                                  float* __restrict__ c) {
  const int lane = threadIdx.x;
  probe_bf16v8 av = *reinterpret_cast<const probe_bf16v8*>(a + lane * 8);
  probe_bf16v8 bv = *reinterpret_cast<const probe_bf16v8*>(b + lane * 8);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[lane * 4 + r] = acc[r];
}
}  // namespace

void mfma_probe(torch::Tensor c, torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.numel() == 512 && b.numel() == 512 && c.numel() == 256);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     c.data_ptr<float>());
}
