// Paged GQA decode attention (gfx950) — split-KV two-kernel design.
//
// Serves the concurrent-agent decode step (SURVEY.md §2.3 "Decode attention
// kernel" row; BASELINE.json configs 2-5). HBM-bound: the whole op streams
// seq_len * 2 * D bf16 of K/V per (seq, kv-head) ONCE — the GQA group's
// `ratio` query heads share every K/V byte loaded.
//
// Kernel 1 (partials): grid (n_seqs, n_kv, SPLITS), ONE wave each.
//   A wave owns a chunk-aligned KV span and ALL `ratio` q heads of its
//   kv head (RATIO is a template constant so the per-head accumulators
//   stay in registers — runtime indexing would spill to scratch).
//     score phase: lane = token. K layout [page][h][D/8][ps][8] gives
//       16-lane-contiguous 16 B loads; each loaded k-vector feeds RATIO
//       dot products against q rows broadcast from LDS.
//     PV phase: lane = dim pair; V rows read once (4 B/lane contiguous),
//       each value feeds RATIO accumulators via p broadcast from LDS.
//   Online softmax per span; unnormalized (m, l, o) partials to workspace.
// Kernel 2 (combine): grid (n_seqs, n_q), one wave: merge the SPLITS
//   partials with the standard log-sum-exp rescale, normalize, write bf16.
//
// Parallelism: B*n_kv*SPLITS waves (e.g. 64 seqs x 8 kv x 8 = 4096 waves
// on 256 CUs) — the v1 single-kernel design peaked at 2 workgroups/CU and
// 0.36 TB/s; this shape exists to keep every CU's memory queue full.
//
// Workspace layout: part[b][g][split][h][PART_STRIDE] f32 where
// [0..127] = o pairs (lane d), [128] = m, [129] = l.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cfloat>
#include "common.h"

namespace {

constexpr int CHUNK = 64;
constexpr int SPLITS = 8;
constexpr int PART_STRIDE = 132;  // 128 o + m + l (+2 pad)

// CT = short (bf16 pages) or unsigned char (fp8 e4m3 pages, scale-free):
// the fp8 path reads HALF the KV bytes and pays VALU converts instead of
// v_dot2 — a good trade everywhere the kernel is HBM- or latency-bound.
template <int RATIO, int D, typename CT>
__global__ __launch_bounds__(64) void decode_partial_kernel(
    float* __restrict__ part,           // [B, n_kv, SPLITS, RATIO, PART_STRIDE]
    const short* __restrict__ q,        // [B, n_q, D]
    const CT* __restrict__ k_cache,     // [P, n_kv, D/8, PS, 8]
    const CT* __restrict__ v_cache,     // [P, n_kv, PS, D]
    const int* __restrict__ page_table, // [B, max_pages]
    const int* __restrict__ seq_lens,   // [B]
    float scale, int n_q, int n_kv, int PS, int max_pages, long q_ts) {
  constexpr bool FP8 = sizeof(CT) == 1;
  const int b = blockIdx.x;
  const int g = blockIdx.y;
  const int split = blockIdx.z;
  const int lane = threadIdx.x;
  const int len = seq_lens[b];
  constexpr int D8 = D / 8;

  float* my_part = part +
      ((((long)b * gridDim.y + g) * SPLITS + split) * RATIO) * PART_STRIDE;

  const int n_chunks = (len + CHUNK - 1) / CHUNK;
  const int cpw = (n_chunks + SPLITS - 1) / SPLITS;  // chunks per wave
  const int c0 = split * cpw;
  const int c1 = min(c0 + cpw, n_chunks);
  if (len <= 0 || c0 >= c1) {
    // empty span: publish neutral partials
    for (int h = 0; h < RATIO; ++h) {
      float* pp = my_part + h * PART_STRIDE;
      pp[2 * lane] = 0.f;
      pp[2 * lane + 1] = 0.f;
      if (lane == 0) { pp[128] = -FLT_MAX; pp[129] = 0.f; }
    }
    return;
  }

  __shared__ float p_lds[RATIO][CHUNK];
  const int qh0 = g * RATIO;
  // score-phase lane split: ds = lane/16 owns a 32-dim slice, t16 = lane%16
  // walks tokens. q lives in REGISTERS as bf16 pairs and the dots run on
  // v_dot2_f32_bf16 — the previous LDS-broadcast-per-element formulation
  // was instruction-issue-bound (512 ds_read + 512 fma per token).
  const int ds = lane / 16;
  const int t16 = lane % 16;
  typedef __bf16 bf2 __attribute__((ext_vector_type(2)));
  // lane (ds, t16) owns dims {(4r+ds)*8 .. +8} for r in 0..3: for each r
  // the wave's 64 lanes cover 4 CONSECUTIVE d8 groups x 16 consecutive
  // tokens = one contiguous 1 KB run per page (full coalescing).
  // q parks in LDS (RATIO x 256 B/wave) and is read per k-slice as a
  // broadcast b128 — holding it in VGPRs cost 64 registers (an occupancy
  // level at RATIO=4).
  __shared__ short q_lds[RATIO][D];
#pragma unroll
  for (int h = 0; h < RATIO; ++h) {
    const short* qp = q + (long)b * q_ts + (long)(qh0 + h) * D;
#pragma unroll
    for (int r = 0; r < D / WAVE; ++r)
      q_lds[h][r * WAVE + lane] = qp[r * WAVE + lane];
  }
  __builtin_amdgcn_s_waitcnt(0);

  // fp8 MFMA score phase operands: q quantized per head to e4m3 in LDS
  // (B operand, row stride padded to dodge 16-way bank conflicts) plus
  // the per-head scales. Scores come from v_mfma_f32_16x16x32_fp8_fp8 —
  // CDNA4 has no VALU fp8 dot, and the cvt+fma fallback was VALU-bound
  // (225 us vs bf16's 204 at ctx 4096 despite half the bytes).
  constexpr int Q8S = D + 8;  // bytes per padded q8 row
  __shared__ unsigned char q8_lds[16 * Q8S];
  __shared__ float qsc_lds[16];
  if constexpr (sizeof(CT) == 1) {
#pragma unroll
    for (int h = 0; h < 16; ++h) {
      float sc = 0.f;
      if (h < RATIO) {
        float mx = fmaxf(fabsf(bits2f(q_lds[h][2 * lane])),
                         fabsf(bits2f(q_lds[h][2 * lane + 1])));
        mx = wave_max(mx);
        sc = fmaxf(mx, 1e-8f) / 448.f;
        const float inv = 1.f / sc;
        q8_lds[h * Q8S + 2 * lane] =
            f2fp8(bits2f(q_lds[h][2 * lane]) * inv);
        q8_lds[h * Q8S + 2 * lane + 1] =
            f2fp8(bits2f(q_lds[h][2 * lane + 1]) * inv);
      } else {
        q8_lds[h * Q8S + 2 * lane] = 0;
        q8_lds[h * Q8S + 2 * lane + 1] = 0;
      }
      if (lane == 0) qsc_lds[h] = sc;
    }
    __builtin_amdgcn_s_waitcnt(0);
  }

  // PV lane split: pv_tg = lane/16 walks tokens (4 per u-step, adjacent
  // rows => 1 KB contiguous per load instruction), pv_dg = lane%16 owns
  // 8 dims (one 16 B load per row). o[RATIO][8] keeps the register count
  // inside 3 waves/SIMD.
  const int pv_tg = lane / 16;
  const int pv_dg = lane % 16;
  float m[RATIO], l[RATIO];
  float o[RATIO][8];
#pragma unroll
  for (int h = 0; h < RATIO; ++h) {
    m[h] = -FLT_MAX; l[h] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[h][j] = 0.f;
  }
  const int* pt = page_table + (long)b * max_pages;

  for (int c = c0; c < c1; ++c) {
    if constexpr (FP8) {
      // MFMA score phase: S^T[16 tok x 16 head-col] per 16-token group.
      // A = K rows (lane: tok = l%16, k-dims (l/16)*8..+8 per 32-dim
      // chunk — 16 lanes x 8 B = one contiguous page-row span), B = q8
      // from LDS. C lane l holds tokens g*16 + (l/16)*4 + r of head
      // l%16. Per-head chunk stats cross lanes through small LDS
      // broadcast arrays; the m/l/alpha bookkeeping stays statically
      // indexed (a dynamic m[myh] would spill the accumulators).
      __shared__ float cm_b[16], cs_b[16], mn_b[16];
      const int myh = lane % 16;
      float sv[16];  // this lane's 16 token scores for head `myh`
#pragma unroll
      for (int gt = 0; gt < 4; ++gt) {
        const int tok = c * CHUNK + gt * 16 + (lane % 16);
        const int tok_c = min(tok, len - 1);
        const long page = pt[tok_c / PS];
        const CT* kp = k_cache +
            ((long)page * n_kv + g) * D8 * PS * 8 + (tok_c % PS) * 8;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int dc = 0; dc < 4; ++dc) {
          long a, bq;
          __builtin_memcpy(&a, kp + (long)(dc * 4 + lane / 16) * PS * 8, 8);
          __builtin_memcpy(&bq, &q8_lds[myh * Q8S + dc * 32 + (lane / 16) * 8],
                           8);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, bq, acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) sv[gt * 4 + r] = acc[r];
      }
      // mask + scale, per-head chunk max (lanes sharing l%16: xor 16/32)
      const float qs = qsc_lds[myh] * scale;
      float cmax = -FLT_MAX;
#pragma unroll
      for (int gt = 0; gt < 4; ++gt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int tok = c * CHUNK + gt * 16 + (lane / 16) * 4 + r;
          const float v = (tok < len) ? sv[gt * 4 + r] * qs : -FLT_MAX;
          sv[gt * 4 + r] = v;
          cmax = fmaxf(cmax, v);
        }
      cmax = fmaxf(cmax, __shfl_xor(cmax, 16, WAVE));
      cmax = fmaxf(cmax, __shfl_xor(cmax, 32, WAVE));
      if (lane < 16) cm_b[myh] = cmax;
      __builtin_amdgcn_s_waitcnt(0);
      // statically-indexed running-max update; new maxima broadcast back
      float alpha_s[RATIO];
#pragma unroll
      for (int h = 0; h < RATIO; ++h) {
        const float mn = fmaxf(m[h], cm_b[h]);
        alpha_s[h] = (m[h] == -FLT_MAX) ? 0.f : __expf(m[h] - mn);
#pragma unroll
        for (int j = 0; j < 8; ++j) o[h][j] *= alpha_s[h];
        m[h] = mn;
        if (lane == 0) mn_b[h] = mn;
      }
      __builtin_amdgcn_s_waitcnt(0);
      // p = exp(s - mn) -> p_lds; per-head chunk sum via the same xor net
      const float mn_my = mn_b[myh];
      float csum = 0.f;
#pragma unroll
      for (int gt = 0; gt < 4; ++gt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int t = gt * 16 + (lane / 16) * 4 + r;
          const float pv = (sv[gt * 4 + r] == -FLT_MAX)
                               ? 0.f
                               : __expf(sv[gt * 4 + r] - mn_my);
          if (myh < RATIO) p_lds[myh][t] = pv;
          csum += pv;
        }
      csum += __shfl_xor(csum, 16, WAVE);
      csum += __shfl_xor(csum, 32, WAVE);
      if (lane < 16) cs_b[myh] = csum;
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int h = 0; h < RATIO; ++h) l[h] = l[h] * alpha_s[h] + cs_b[h];
    } else {
    float s[4][RATIO];  // [16-token sub-pass][head]
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      const int tok = c * CHUNK + sub * 16 + t16;
      float acc[RATIO];
#pragma unroll
      for (int h = 0; h < RATIO; ++h) acc[h] = 0.f;
      if (tok < len) {
        const long page = pt[tok / PS];
        // this lane's 4 d8 groups: d8 = r*4 + ds (coalesced per r)
        const CT* kp = k_cache +
            ((long)page * n_kv + g) * D8 * PS * 8 + (tok % PS) * 8;
        {
          bf16x8 kv[4];
#pragma unroll
          for (int r = 0; r < 4; ++r)
            kv[r] = *reinterpret_cast<const bf16x8*>(
                kp + (long)(r * 4 + ds) * PS * 8);
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const bf2* kp2 = reinterpret_cast<const bf2*>(&kv[r]);
            // this lane's dims (4r+ds)*8..+8 of each head's q: broadcast
            // LDS reads (t16 lanes share the address — conflict-free)
#pragma unroll
            for (int h = 0; h < RATIO; ++h) {
              const bf16x8 qv8 = *reinterpret_cast<const bf16x8*>(
                  &q_lds[h][(r * 4 + ds) * 8]);
              const bf2* qp2 = reinterpret_cast<const bf2*>(&qv8);
#pragma unroll
              for (int j = 0; j < 4; ++j)
                acc[h] = __builtin_amdgcn_fdot2_f32_bf16(kp2[j], qp2[j],
                                                         acc[h], false);
            }
          }
        }
      }
      // reduce the 4 dim-slices (lanes differing in bits 4..5)
#pragma unroll
      for (int h = 0; h < RATIO; ++h) {
        acc[h] += __shfl_xor(acc[h], 16, WAVE);
        acc[h] += __shfl_xor(acc[h], 32, WAVE);
        s[sub][h] = (tok < len) ? acc[h] * scale : -FLT_MAX;
      }
    }
    // per-head online softmax update over the whole 64-token chunk
#pragma unroll
    for (int h = 0; h < RATIO; ++h) {
      float cm = fmaxf(fmaxf(s[0][h], s[1][h]), fmaxf(s[2][h], s[3][h]));
      cm = wave_max(cm);
      const float mn = fmaxf(m[h], cm);
      float psum = 0.f;
#pragma unroll
      for (int sub = 0; sub < 4; ++sub) {
        const float p = (s[sub][h] == -FLT_MAX) ? 0.f : __expf(s[sub][h] - mn);
        if (ds == 0) {
          p_lds[h][sub * 16 + t16] = p;
          psum += p;
        }
      }
      const float csum = wave_sum(psum);
      const float alpha = (m[h] == -FLT_MAX) ? 0.f : __expf(m[h] - mn);
      l[h] = l[h] * alpha + csum;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[h][j] *= alpha;
      m[h] = mn;
    }
    __builtin_amdgcn_s_waitcnt(0);
    }

    // PV: per u-step this lane reads V[tok = base + u*4 + pv_tg]
    // [dims pv_dg*8 .. +8) — the wave covers 4 adjacent token rows x
    // full 256-B width = 1 KB contiguous; p broadcasts from LDS.
    const int base_tok = c * CHUNK;
#pragma unroll 4
    for (int u = 0; u < 16; ++u) {
      const int t = u * 4 + pv_tg;
      const int gt = base_tok + t;
      const int gt_c = min(gt, len - 1);
      const long page = pt[gt_c / PS];
      const CT* vp = v_cache +
          (((long)page * n_kv + g) * PS + gt_c % PS) * D + pv_dg * 8;
      float vf[8];
      if constexpr (FP8) {
        const u8x8 vv = *reinterpret_cast<const u8x8*>(vp);
        fp8x8_to_f32(vv, vf);
      } else {
        const bf16x8 vv = *reinterpret_cast<const bf16x8*>(vp);
#pragma unroll
        for (int j = 0; j < 8; ++j) vf[j] = bits2f(vv[j]);
      }
#pragma unroll
      for (int h = 0; h < RATIO; ++h) {
        const float pw = p_lds[h][t];  // 0 beyond len
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[h][j] = fmaf(pw, vf[j], o[h][j]);
      }
    }
  }

  // publish partials (unnormalized): reduce o across the 4 token groups
  // (lane bits 4..5), then pv_tg==0 lanes own dims [pv_dg*8, pv_dg*8+8)
#pragma unroll
  for (int h = 0; h < RATIO; ++h) {
    float* pp = my_part + h * PART_STRIDE;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = o[h][j];
      v += __shfl_xor(v, 16, WAVE);
      v += __shfl_xor(v, 32, WAVE);
      if (pv_tg == 0) pp[pv_dg * 8 + j] = v;
    }
    if (lane == 0) { pp[128] = m[h]; pp[129] = l[h]; }
  }
}

template <int D>
__global__ __launch_bounds__(64) void decode_combine_kernel(
    short* __restrict__ out,          // [B, n_q, D]
    const float* __restrict__ part,   // [B, n_kv, SPLITS, RATIO, PART_STRIDE]
    const int* __restrict__ seq_lens, int n_q, int n_kv) {
  const int b = blockIdx.x;
  const int qh = blockIdx.y;
  const int lane = threadIdx.x;
  if (seq_lens[b] <= 0) return;
  const int ratio = n_q / n_kv;
  const int g = qh / ratio;
  const int h = qh % ratio;
  const float* pp = part +
      ((((long)b * n_kv + g) * SPLITS) * ratio + h) * PART_STRIDE;
  const long step = (long)ratio * PART_STRIDE;

  float M = -FLT_MAX;
#pragma unroll
  for (int s = 0; s < SPLITS; ++s) M = fmaxf(M, pp[s * step + 128]);
  float L = 0.f, O0 = 0.f, O1 = 0.f;
#pragma unroll
  for (int s = 0; s < SPLITS; ++s) {
    const float ms = pp[s * step + 128];
    if (ms == -FLT_MAX) continue;
    const float w = __expf(ms - M);
    L += w * pp[s * step + 129];
    O0 = fmaf(w, pp[s * step + 2 * lane], O0);
    O1 = fmaf(w, pp[s * step + 2 * lane + 1], O1);
  }
  const float inv = (L > 0.f) ? 1.0f / L : 0.f;
  short* op = out + ((long)b * n_q + qh) * D + 2 * lane;
  op[0] = f2bits(O0 * inv);
  op[1] = f2bits(O1 * inv);
}

}  // namespace

void paged_decode_attention_ws(torch::Tensor out, torch::Tensor q,
                               torch::Tensor k_cache, torch::Tensor v_cache,
                               torch::Tensor page_table, torch::Tensor seq_lens,
                               torch::Tensor workspace, double scale) {
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(page_table.scalar_type() == at::kInt &&
              seq_lens.scalar_type() == at::kInt);
  TORCH_CHECK(workspace.scalar_type() == at::kFloat);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  const int B = q.size(0), n_q = q.size(1), D = q.size(2);
  const int n_kv = k_cache.size(1), PS = k_cache.size(3);
  const int max_pages = page_table.size(1);
  const int ratio = n_q / n_kv;
  TORCH_CHECK(D == 128, "decode attention: head_dim 128 only");
  TORCH_CHECK(n_q % n_kv == 0);
  TORCH_CHECK(workspace.numel() >=
              (long)B * n_kv * SPLITS * ratio * PART_STRIDE,
              "decode workspace too small");
  if (B == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid1(B, n_kv, SPLITS);
  const bool fp8 = k_cache.scalar_type() == at::kFloat8_e4m3fn;
#define LAUNCH_RATIO_CT(R, CT)                                                 \
  hipLaunchKernelGGL((decode_partial_kernel<R, 128, CT>), grid1, dim3(64), 0,  \
                     stream, workspace.data_ptr<float>(),                      \
                     (const short*)q.data_ptr(),                               \
                     (const CT*)k_cache.data_ptr(),                            \
                     (const CT*)v_cache.data_ptr(),                            \
                     page_table.data_ptr<int>(), seq_lens.data_ptr<int>(),     \
                     (float)scale, n_q, n_kv, PS, max_pages,                    \
                     (long)q.stride(0))
#define LAUNCH_RATIO(R)                                                        \
  do {                                                                         \
    if (fp8) LAUNCH_RATIO_CT(R, unsigned char);                                \
    else LAUNCH_RATIO_CT(R, short);                                            \
  } while (0)
  switch (ratio) {
    case 1: LAUNCH_RATIO(1); break;
    case 2: LAUNCH_RATIO(2); break;
    case 4: LAUNCH_RATIO(4); break;
    case 8: LAUNCH_RATIO(8); break;
    default:
      TORCH_CHECK(false, "decode attention: GQA ratio must be 1/2/4/8, got ",
                  ratio);
  }
#undef LAUNCH_RATIO
#undef LAUNCH_RATIO_CT
  hipLaunchKernelGGL((decode_combine_kernel<128>), dim3(B, n_q), dim3(64), 0,
                     stream, (short*)out.data_ptr(),
                     workspace.data_ptr<float>(), seq_lens.data_ptr<int>(),
                     n_q, n_kv);
}

// Workspace floats needed for (B, n_q/n_kv ratio, n_kv).
long decode_attention_workspace_size(long B, long n_q, long n_kv) {
  return B * n_kv * SPLITS * (n_q / n_kv) * PART_STRIDE;
}

void paged_decode_attention(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor page_table, torch::Tensor seq_lens,
                            double scale) {
  const int B = q.size(0), n_q = q.size(1);
  const int n_kv = k_cache.size(1);
  auto ws = torch::empty(
      {decode_attention_workspace_size(B, n_q, n_kv)},
      torch::TensorOptions().dtype(torch::kFloat).device(q.device()));
  paged_decode_attention_ws(out, q, k_cache, v_cache, page_table, seq_lens,
                            ws, scale);
}
