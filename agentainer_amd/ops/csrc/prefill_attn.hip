// Paged varlen causal prefill attention (gfx950, MFMA bf16).
//
// Flash-style online-softmax attention for the batched prefill phase of
// the continuous-batching scheduler (SURVEY.md §2.3 "Prefill attention
// kernel" row). Q rows are the batch's NEW tokens (varlen-packed); K/V are
// read from the paged pools, so a conversation's cached prefix is attended
// without re-prefilling it (the agent-KV-persistence contract).
//
// Geometry (one workgroup = 4 waves = one 32-row Q tile of 4 GQA heads):
//   grid: (ceil(max_qlen/32), n_seqs, n_q/4); wave w -> head z*4+w.
//   All 4 heads share one kv head (requires ratio % 4 == 0 — true for
//   Llama-3 8B/70B and Mixtral), so the K/V tiles staged in LDS are shared.
//   KV tile = 32 tokens. Per tile, per wave:
//     S^T tiles  = mfma_f32_16x16x32_bf16(A=Q, B=K)   [16 q x 16 kv] x 2x2
//     P -> LDS (bf16), online m/l update per q row
//     O tiles   += mfma(A=P, B=V^T)                    [16 q x 16 d] x 2x8
//   MFMA fragment maps (gfx950): A lane l holds A[l%16][(l/16)*8+j];
//   B lane l holds B[(l/16)*8+j][l%16]; C/D lane l holds rows (l/16)*4+r,
//   col l%16 (cdna_hip_programming.md §3).
//
// LDS: K tile [32][128] XOR-swizzled (row-major D=128 is a 16-way bank
// conflict for b128 reads otherwise — G4), V tile stored transposed
// [128][32+8], per-wave P buffers. Single-buffered v1: stage -> barrier ->
// compute -> barrier.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cfloat>
#include "common.h"

namespace {

typedef __bf16 bf16v8 __attribute__((ext_vector_type(8)));

constexpr int QBLK = 32;
constexpr int KVBLK = 32;
constexpr int D = 128;
constexpr int VT_STRIDE = KVBLK + 8;   // shorts; keeps 16-B alignment
constexpr int P_STRIDE = KVBLK + 8;

__device__ __forceinline__ int kswz(int tok, int byte_off) {
  // XOR swizzle within a K row: spread the 16-B slot by token (G4/T2)
  return byte_off ^ ((tok & 7) << 4);
}

// CT = short (bf16 pages) or unsigned char (fp8 e4m3 pages): the fp8
// path dequantizes to bf16 during LDS staging, so the MFMA pipeline and
// both LDS images are byte-identical to the bf16 build.
template <typename CT>
__device__ __forceinline__ bf16x8 load8_cache(const CT* p) {
  if constexpr (sizeof(CT) == 1) {
    const u8x8 b = *reinterpret_cast<const u8x8*>(p);
    float f[8];
    fp8x8_to_f32(b, f);
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) r[j] = f2bits(f[j]);
    return r;
  } else {
    return *reinterpret_cast<const bf16x8*>(p);
  }
}

template <typename CT>
__global__ __launch_bounds__(256, 1) void prefill_attn_kernel(
    short* __restrict__ out,            // [Tq, n_q, D]
    const short* __restrict__ q,        // [Tq, n_q, D]
    const CT* __restrict__ k_cache,     // [P, n_kv, D/8, PS, 8]
    const CT* __restrict__ v_cache,     // [P, n_kv, PS, D]
    const int* __restrict__ page_table, // [B, max_pages]
    const int* __restrict__ seq_lens,   // [B] total context length
    const int* __restrict__ q_starts,   // [B] row offset into q
    const int* __restrict__ q_lens,     // [B]
    float scale, int n_q, int n_kv, int PS, int max_pages, long q_ts) {
  const int qtile = blockIdx.x;
  const int b = blockIdx.y;
  const int head = blockIdx.z * 4 + threadIdx.x / WAVE;
  const int qlen = q_lens[b];
  if (qtile * QBLK >= qlen) return;
  const int seq_len = seq_lens[b];
  const int ctx_start = seq_len - qlen;  // new tokens sit at the end
  const int ratio = n_q / n_kv;
  const int g = head / ratio;
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l16 = lane % 16;    // fragment col / A row
  const int lg = lane / 16;     // fragment k-group

  __shared__ short k_lds[2][QBLK * D];         // swizzled [32][128], dbuf
  __shared__ short vt_lds[2][D * VT_STRIDE];   // [128][32+8], dbuf
  __shared__ short p_lds[4][16 * P_STRIDE];    // per wave [16 q][32+8 kv]

  const int* pt = page_table + (long)b * max_pages;
  const long q_row0 = q_starts[b];

  // ---- load Q fragments: [2 qsub][4 dchunk], lane l -> Q[qsub*16+l16][dc*32+lg*8 ..+8]
  bf16v8 qf[2][4];
#pragma unroll
  for (int qs = 0; qs < 2; ++qs) {
    const int qr = qtile * QBLK + qs * 16 + l16;
#pragma unroll
    for (int dc = 0; dc < 4; ++dc) {
      if (qr < qlen) {
        const short* p = q + (q_row0 + qr) * q_ts + (long)head * D + dc * 32 + lg * 8;
        qf[qs][dc] = *reinterpret_cast<const bf16v8*>(p);
      } else {
        qf[qs][dc] = bf16v8{};
      }
    }
  }

  // ---- accumulators
  f32x4 o[2][8];  // [qsub][dim tile]
#pragma unroll
  for (int qs = 0; qs < 2; ++qs)
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) o[qs][dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int qs = 0; qs < 2; ++qs)
#pragma unroll
    for (int r = 0; r < 4; ++r) { m_run[qs][r] = -FLT_MAX; l_run[qs][r] = 0.f; }

  const int max_qpos = ctx_start + min(qtile * QBLK + QBLK - 1, qlen - 1);
  const int kv_end = max_qpos + 1;
  const int n_tiles = (kv_end + KVBLK - 1) / KVBLK;

  // T14 split staging: each thread owns 2 K chunks + 2 V chunks of a tile;
  // tile t+1's GLOBAL loads issue before tile t's MFMAs, the LDS writes land
  // after them, ONE barrier per tile.
  const int kc0 = threadIdx.x;            // K chunk ids (tok = c%32, d8 = c/32)
  const int kc1 = threadIdx.x + 256;
  bf16x8 st_k0, st_k1, st_v0, st_v1;
  auto stage_load = [&](int t) {
    const int kv0 = t * KVBLK;
    auto loadk = [&](int c) -> bf16x8 {
      const int tok = c % QBLK, d8 = c / QBLK;
      const int gt = kv0 + tok;
      bf16x8 vvv{};
      if (gt < kv_end && gt < seq_len) {
        const long pg = pt[gt / PS];
        vvv = load8_cache<CT>(
            k_cache + ((((pg * n_kv + g) * (D / 8) + d8) * PS) + gt % PS) * 8);
      }
      return vvv;
    };
    auto loadv = [&](int c) -> bf16x8 {
      const int tok = c / 16, d0 = (c % 16) * 8;
      const int gt = kv0 + tok;
      bf16x8 vvv{};
      if (gt < kv_end && gt < seq_len) {
        const long pg = pt[gt / PS];
        vvv = load8_cache<CT>(
            v_cache + (((pg * n_kv + g) * PS) + gt % PS) * D + d0);
      }
      return vvv;
    };
    st_k0 = loadk(kc0);
    st_k1 = loadk(kc1);
    st_v0 = loadv(kc0);
    st_v1 = loadv(kc1);
  };
  auto stage_write = [&](int buf) {
    auto writek = [&](int c, bf16x8 vvv) {
      const int tok = c % QBLK, d8 = c / QBLK;
      const int byte0 = tok * (D * 2) + d8 * 16;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(k_lds[buf]) + kswz(tok, byte0)) = vvv;
    };
    auto writev = [&](int c, bf16x8 vvv) {
      const int tok = c / 16, d0 = (c % 16) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[buf][(d0 + j) * VT_STRIDE + tok] = vvv[j];
    };
    writek(kc0, st_k0);
    writek(kc1, st_k1);
    writev(kc0, st_v0);
    writev(kc1, st_v1);
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  int buf = 0;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * KVBLK;
    const bool has_next = (t + 1 < n_tiles);
    if (has_next) stage_load(t + 1);  // next tile's HBM loads in flight

    // ---- K fragments for this tile: [2 ksub][4 dchunk]
    bf16v8 kf[2][4];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int tok = ks * 16 + l16;
#pragma unroll
      for (int dc = 0; dc < 4; ++dc) {
        const int byte0 = tok * (D * 2) + (dc * 32 + lg * 8) * 2;
        kf[ks][dc] = *reinterpret_cast<const bf16v8*>(
            reinterpret_cast<const char*>(k_lds[buf]) + kswz(tok, byte0));
      }
    }
    // ---- V^T fragments: [8 dim tiles]
    bf16v8 vf[8];
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      const int dim = dt * 16 + l16;
      vf[dt] = *reinterpret_cast<const bf16v8*>(
          &vt_lds[buf][dim * VT_STRIDE + lg * 8]);
    }

#pragma unroll
    for (int qs = 0; qs < 2; ++qs) {
      // S^T tiles: [2 ksub][16q x 16kv]
      f32x4 s[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        s[ks] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int dc = 0; dc < 4; ++dc)
          s[ks] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[qs][dc], kf[ks][dc], s[ks], 0, 0, 0);
      }
      // mask + row stats. lane holds rows (lg*4+r), col l16 (kv).
      float rmax[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qr = qtile * QBLK + qs * 16 + lg * 4 + r;
        const int qpos = ctx_start + qr;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int kpos = kv0 + ks * 16 + l16;
          float v = s[ks][r] * scale;
          if (qr >= qlen || kpos > qpos) v = -FLT_MAX;
          s[ks][r] = v;
        }
        float mx = fmaxf(s[0][r], s[1][r]);
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        rmax[r] = mx;
      }
      // online update
      float alpha[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float mn = fmaxf(m_run[qs][r], rmax[r]);
        alpha[r] = (m_run[qs][r] == -FLT_MAX) ? 0.f : __expf(m_run[qs][r] - mn);
        m_run[qs][r] = mn;
        float rowsum = 0.f;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          float p = (s[ks][r] == -FLT_MAX) ? 0.f : __expf(s[ks][r] - mn);
          s[ks][r] = p;
          rowsum += p;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          rowsum += __shfl_xor(rowsum, off, WAVE);
        l_run[qs][r] = l_run[qs][r] * alpha[r] + rowsum;
      }
      // P -> LDS (bf16) for the A-fragment of PV
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wid][(lg * 4 + r) * P_STRIDE + ks * 16 + l16] = f2bits(s[ks][r]);
      __builtin_amdgcn_s_waitcnt(0);  // own-wave LDS ordering
      // rescale O
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
#pragma unroll
        for (int r = 0; r < 4; ++r) o[qs][dt][r] *= alpha[r];
      // PV: A = P fragment (lane: P[l16][lg*8+j]), B = V^T fragment
      const bf16v8 pf = *reinterpret_cast<const bf16v8*>(
          &p_lds[wid][l16 * P_STRIDE + lg * 8]);
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        o[qs][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, vf[dt], o[qs][dt], 0, 0, 0);
    }
    if (has_next) stage_write(buf ^ 1);  // writes land after the MFMAs
    __syncthreads();
    buf ^= 1;
  }

  // ---- epilogue: O /= l, write bf16
#pragma unroll
  for (int qs = 0; qs < 2; ++qs) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qr = qtile * QBLK + qs * 16 + lg * 4 + r;
      if (qr >= qlen) continue;
      const float inv = (l_run[qs][r] > 0.f) ? 1.0f / l_run[qs][r] : 0.f;
      short* op = out + ((q_row0 + qr) * n_q + head) * D;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        op[dt * 16 + l16] = f2bits(o[qs][dt][r] * inv);
    }
  }
}

}  // namespace

void paged_prefill_attention(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor page_table, torch::Tensor seq_lens,
                             torch::Tensor query_starts, torch::Tensor query_lens,
                             double scale, long max_qlen) {
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  const int n_q = q.size(1), Dh = q.size(2);
  const int n_kv = k_cache.size(1), PS = k_cache.size(3);
  const int B = seq_lens.size(0);
  const int max_pages = page_table.size(1);
  TORCH_CHECK(Dh == 128, "prefill attention: head_dim 128 only");
  TORCH_CHECK(n_q % 4 == 0, "n_q must be a multiple of 4");
  const int ratio = n_q / n_kv;
  TORCH_CHECK(ratio % 4 == 0 || ratio == n_q,
              "GQA ratio must be a multiple of 4 (waves of a workgroup share "
              "one kv head)");
  if (max_qlen <= 0) {
    // fallback: derive from the device tensor. This D2H copy BLOCKS the
    // host until the stream drains (measured ~2 ms/call under serving
    // load with a deep queue — 32x per prefill); hot callers pass
    // max_qlen explicitly (AttnMetadata.max_qlen).
    auto q_lens_cpu = query_lens.to(torch::kCPU);
    auto* ql = q_lens_cpu.data_ptr<int>();
    for (int i = 0; i < B; ++i)
      max_qlen = std::max<long>(max_qlen, ql[i]);
    if (max_qlen == 0) return;
  }
  const int qtiles = (max_qlen + QBLK - 1) / QBLK;
  auto stream = at::hip::getCurrentHIPStream();
#define PF_LAUNCH(CT)                                                          \
  hipLaunchKernelGGL(prefill_attn_kernel<CT>, dim3(qtiles, B, n_q / 4),        \
                     dim3(256), 0, stream, (short*)out.data_ptr(),             \
                     (const short*)q.data_ptr(),                               \
                     (const CT*)k_cache.data_ptr(),                            \
                     (const CT*)v_cache.data_ptr(),                            \
                     page_table.data_ptr<int>(), seq_lens.data_ptr<int>(),     \
                     query_starts.data_ptr<int>(), query_lens.data_ptr<int>(), \
                     (float)scale, n_q, n_kv, PS, max_pages,                    \
                     (long)q.stride(0))
  if (k_cache.scalar_type() == at::kFloat8_e4m3fn) {
    PF_LAUNCH(unsigned char);
  } else {
    TORCH_CHECK(k_cache.scalar_type() == at::kBFloat16);
    PF_LAUNCH(short);
  }
#undef PF_LAUNCH
}
