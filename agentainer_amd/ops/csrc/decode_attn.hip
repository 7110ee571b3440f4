// Paged GQA decode attention (gfx950) — one new token per sequence.
//
// Serves the concurrent-agent decode step (SURVEY.md §2.3 "Decode attention
// kernel" row; BASELINE.json configs 2-5). The op is HBM-bound: each
// (sequence, kv-head) streams seq_len * 2 * D bf16 of K/V once. Design:
//
//   grid:  (n_seqs, n_kv_heads); block: 4 waves (256 threads).
//   Each wave owns one query head of the GQA group (ratio > 4: waves loop).
//   KV is consumed in 64-token chunks with two lane roles:
//     score phase: lane = token. K layout [page][h][D/8][ps][8] makes the
//       16-lane page subgroups read 16 B contiguous per (d8, token) — fully
//       coalesced. q (f32) lives in LDS and broadcasts (all lanes read the
//       same address = LDS broadcast, conflict-free).
//     PV phase:   lane = dim pair. V layout [page][h][ps][D] makes lanes
//       read 4 B contiguous across a token row. p broadcasts from LDS.
//   Online softmax (m, l running; rescale the 2-f32 o accumulator).
//
// Wave-shape notes: all reductions are 64-wide __shfl_xor (wave64, never
// warp-32 idioms); per-wave LDS slices avoid __syncthreads in the KV loop
// (s_waitcnt lgkmcnt orders each wave's own LDS accesses).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cfloat>
#include "common.h"

namespace {

constexpr int CHUNK = 64;

template <int D>
__global__ __launch_bounds__(256) void decode_attn_kernel(
    short* __restrict__ out,            // [B, n_q, D]
    const short* __restrict__ q,        // [B, n_q, D]
    const short* __restrict__ k_cache,  // [P, n_kv, D/8, PS, 8]
    const short* __restrict__ v_cache,  // [P, n_kv, PS, D]
    const int* __restrict__ page_table, // [B, max_pages]
    const int* __restrict__ seq_lens,   // [B]
    float scale, int n_q, int n_kv, int PS, int max_pages, int ratio) {
  const int b = blockIdx.x;
  const int g = blockIdx.y;  // kv head
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int len = seq_lens[b];
  if (len <= 0) return;
  constexpr int D8 = D / 8;

  // LDS: per-wave q (f32[D]) and p (f32[CHUNK]) slices.
  __shared__ float q_lds[4][D];
  __shared__ float p_lds[4][CHUNK];

  const int* pt = page_table + (long)b * max_pages;
  const int n_chunks = (len + CHUNK - 1) / CHUNK;

  for (int qh = g * ratio + wid; qh < (g + 1) * ratio; qh += 4) {
    // load q for this head into LDS as f32
    const short* qp = q + ((long)b * n_q + qh) * D;
#pragma unroll
    for (int r = 0; r < D / WAVE; ++r)
      q_lds[wid][r * WAVE + lane] = bits2f(qp[r * WAVE + lane]);
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): q_lds visible to this wave

    float m = -FLT_MAX, l = 0.f;
    float o0 = 0.f, o1 = 0.f;  // this lane's two output dims
    const int d0 = lane * 2;

    for (int c = 0; c < n_chunks; ++c) {
      const int tok = c * CHUNK + lane;  // this lane's token (score phase)
      float s = -FLT_MAX;
      if (tok < len) {
        const long page = pt[tok / PS];
        const int off = tok % PS;
        const short* kp = k_cache + (((long)page * n_kv + g) * D8 * PS + off) * 8;
        float acc = 0.f;
#pragma unroll
        for (int d8 = 0; d8 < D8; ++d8) {
          bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(kp + (long)d8 * PS * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc = fmaf(bits2f(kv8[j]), q_lds[wid][d8 * 8 + j], acc);
        }
        s = acc * scale;
      }
      // online softmax update across the wave
      const float cmax = wave_max(s);
      const float mn = fmaxf(m, cmax);
      const float p = (tok < len) ? __expf(s - mn) : 0.f;
      const float csum = wave_sum(p);
      const float alpha = (m == -FLT_MAX) ? 0.f : __expf(m - mn);
      l = l * alpha + csum;
      o0 *= alpha;
      o1 *= alpha;
      m = mn;
      p_lds[wid][lane] = p;
      __builtin_amdgcn_s_waitcnt(0);

      // PV phase: lane = dim pair d0, d0+1
      const int c_len = min(CHUNK, len - c * CHUNK);
      for (int t = 0; t < c_len; ++t) {
        const int gt = c * CHUNK + t;
        const long page = pt[gt / PS];
        const int off = gt % PS;
        const short* vp = v_cache + (((long)page * n_kv + g) * PS + off) * D + d0;
        const float pw = p_lds[wid][t];
        o0 = fmaf(pw, bits2f(vp[0]), o0);
        o1 = fmaf(pw, bits2f(vp[1]), o1);
      }
    }
    const float inv = (l > 0.f) ? 1.0f / l : 0.f;
    short* op = out + ((long)b * n_q + qh) * D + d0;
    op[0] = f2bits(o0 * inv);
    op[1] = f2bits(o1 * inv);
  }
}

}  // namespace

void paged_decode_attention(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor page_table, torch::Tensor seq_lens,
                            double scale) {
  TORCH_CHECK(q.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(page_table.scalar_type() == at::kInt &&
              seq_lens.scalar_type() == at::kInt);
  const int B = q.size(0), n_q = q.size(1), D = q.size(2);
  const int n_kv = k_cache.size(1), PS = k_cache.size(3);
  const int max_pages = page_table.size(1);
  const int ratio = n_q / n_kv;
  TORCH_CHECK(D == 128, "decode attention: head_dim 128 only");
  TORCH_CHECK(n_q % n_kv == 0);
  if (B == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((decode_attn_kernel<128>), dim3(B, n_kv), dim3(256), 0,
                     stream, (short*)out.data_ptr(), (const short*)q.data_ptr(),
                     (const short*)k_cache.data_ptr(),
                     (const short*)v_cache.data_ptr(),
                     page_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                     (float)scale, n_q, n_kv, PS, max_pages, ratio);
}
