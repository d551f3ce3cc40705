// GQA decode attention for CDNA4 (gfx950), flash-decoding style.
//
// Default path (mfma_decode_kernel): the GQ query heads of one kv head form
// the rows of an mfma_f32_16x16x32_bf16 A-fragment (padded to 16), so 32
// cache keys cost 4 QK MFMAs + 8 PV MFMAs per wave instead of 32*GQ
// shuffle-reduced dot products. K is consumed directly from global as
// B-fragments (16 B/lane); V goes through the XOR-swizzled LDS transpose
// (same as the prefill kernel); 4 waves take interleaved 32-key tiles of the
// workgroup's split range and combine through LDS; attn_decode_combine merges
// splits. Measured (scripts/decode_mfma.hip, llama-2-70b shape): 1039 GB/s at
// kv=4k -> 3509 GB/s at kv=131k, 2.2-2.4x the VALU kernel at every length.
//
// The original VALU kernel (attn_decode_kernel: 16-lane-group dot products,
// online softmax in registers) is kept behind PETALS_DECODE_KERNEL=valu.
// kv_len comes from a DEVICE pointer so the whole decode step can be captured
// in a hipGraph with a moving position.
//
// Replaces the reference's torch QK^T/softmax/AV decode math
// (reference models/llama/block.py:108-127) on the MI355X fast path.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdlib>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define DKVT 32    // keys per wave-tile (MFMA path)
#define DWAVES 4   // waves per workgroup (MFMA path)
#define DKPAD 8

#define DVT_BYTE(dim, key_byte) \
  ((((unsigned)(dim)) * ((DKVT + DKPAD) * 2) + (unsigned)(key_byte)) ^ ((((unsigned)(dim) >> 3) & 7u) << 4))

// Arbitrary GQ: heads are processed in ceil(GQ/16) groups of <=16 (one
// A-fragment each). The group loop sits OUTSIDE the kv loop so live registers
// stay at one group's footprint; K/V re-reads per extra group come from
// L2/HBM and are negligible for the weight-bound decode shapes that need
// GQ > 16 (falcon-7b MQA gq=71, falcon-180b gq=29 — reference
// models/falcon/block.py:100-110).
template <int HD, int GQ>
__global__ __launch_bounds__(DWAVES * 64) void mfma_decode_kernel(
    const float* __restrict__ q,            // [B, KV, GQ, HD]
    const unsigned short* __restrict__ k_cache,  // [B, KV, lmax, HD]
    const unsigned short* __restrict__ v_cache,
    float* __restrict__ part_o,             // [B*KV, splits, GQ, HD]
    float* __restrict__ part_ml,            // [B*KV, splits, GQ, 2]
    const int* __restrict__ kv_len_ptr,
    const float* __restrict__ alibi,        // [kv_heads*GQ] slopes or null
    int kv_heads,
    int lmax,
    int n_splits,
    float scale,
    float* __restrict__ out_direct) {  // n_splits==1: normalized out [B, KV*GQ*HD]
                                       // written here; the combine kernel is skipped
  const int bkv = blockIdx.x;
  const int kvh = bkv % kv_heads;
  const int split = blockIdx.y;
  const int kv_len = *kv_len_ptr;

  const int rows_per_split = (kv_len + n_splits - 1) / n_splits;
  const int j_begin = split * rows_per_split;
  const int j_end = min(j_begin + rows_per_split, kv_len);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int hi = lane >> 4;

  constexpr int KCH = HD / 32;
  constexpr int GA = (GQ + 15) / 16;  // head groups of <=16

  // per-wave LDS: swizzled V^T tile + P scratch; the cross-wave combine
  // overlays the V^T storage after the main loop (one <=16-head group at a
  // time, so the overlay always fits)
  __shared__ unsigned char vt_raw[DWAVES][HD * (DKVT + DKPAD) * 2];
  __shared__ unsigned short p_lds[DWAVES][16][DKVT + DKPAD];
  __shared__ float c_ml[DWAVES][16][2];
  static_assert(16 * HD * 4 <= HD * (DKVT + DKPAD) * 2, "combine O overlay too big");

  const size_t kv_base = (size_t)bkv * lmax * HD;
  const unsigned short* kb = k_cache + kv_base;
  const unsigned short* vb = v_cache + kv_base;
  const size_t q_base = (size_t)bkv * GQ * HD;

  for (int a = 0; a < GA; ++a) {
    const int g0 = a * 16;  // first head of this group
    if (a > 0) __syncthreads();  // wave0's combine reads of vt_raw are done

    // ---- q tile: A-fragment rows = q heads of this group (zero-padded), pre-scaled
    bf16x8 q_frag[KCH];
#pragma unroll
    for (int kc = 0; kc < KCH; ++kc) {
      if (g0 + col < GQ) {
        const float* src = q + q_base + (size_t)(g0 + col) * HD + kc * 32 + hi * 8;
        short v[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = (short)f32_to_bf16(src[e] * scale);
        q_frag[kc] = bf16x8{v[0], v[1], v[2], v[3], v[4], v[5], v[6], v[7]};
      } else {
        q_frag[kc] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }

    // per-lane-row ALiBi slopes (rows = heads g0 + hi*4 + r)
    float sl[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int g = g0 + hi * 4 + r;
      sl[r] = (alibi && g < GQ) ? alibi[kvh * GQ + g] : 0.f;
    }

    f32x4 acc_o[HD / 16];
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) acc_o[d] = f32x4{0.f, 0.f, 0.f, 0.f};
    float m_row[4], l_row[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_row[r] = NEG_SENTINEL;
      l_row[r] = 0.f;
    }

    // wave w handles tiles j_begin + (t*DWAVES + w)*DKVT
    for (int j0 = j_begin + wave * DKVT; j0 < j_end; j0 += DWAVES * DKVT) {
      const int tile_n = min(DKVT, j_end - j0);
      const bool full = tile_n == DKVT;

      // ---- stage V^T (this wave only: no block barrier; the waitcnt orders
      // this wave's LDS writes before its reads)
      for (int idx = lane; idx < DKVT * (HD / 8); idx += WAVE) {
        const int row = idx / (HD / 8);
        const int c8 = (idx - row * (HD / 8)) * 8;
        bf16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        if (j0 + row < j_end) vv8 = *reinterpret_cast<const bf16x8*>(vb + (size_t)(j0 + row) * HD + c8);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          *reinterpret_cast<unsigned short*>(&vt_raw[wave][DVT_BYTE(c8 + e, row * 2)]) = (unsigned short)vv8[e];
      }
      __builtin_amdgcn_s_waitcnt(0);

      // ---- S = Q K^T, K direct from global: B[k = kdim][n = key]
      f32x4 s_acc[DKVT / 16];
#pragma unroll
      for (int nb = 0; nb < DKVT / 16; ++nb) {
        s_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};
        const int key = j0 + nb * 16 + col;
        const unsigned short* krow = kb + (size_t)min(key, j_end - 1) * HD + hi * 8;
#pragma unroll
        for (int kc = 0; kc < KCH; ++kc) {
          const bf16x8 kt = *reinterpret_cast<const bf16x8*>(krow + kc * 32);
          s_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[kc], kt, s_acc[nb], 0, 0, 0);
        }
      }

      // ---- online softmax over this tile (rows = heads g0 + hi*4+r)
      float p[DKVT / 16][4];
      float corr[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s[DKVT / 16];
#pragma unroll
        for (int nb = 0; nb < DKVT / 16; ++nb) {
          const int key = j0 + nb * 16 + col;
          s[nb] = s_acc[nb][r] + sl[r] * key;
          if (!full && key >= j_end) s[nb] = NEG_SENTINEL;
        }
        float mx = NEG_SENTINEL;
#pragma unroll
        for (int nb = 0; nb < DKVT / 16; ++nb) mx = fmaxf(mx, s[nb]);
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_row[r], mx);
        corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
        float lsum = 0.f;
#pragma unroll
        for (int nb = 0; nb < DKVT / 16; ++nb) {
          p[nb][r] = (s[nb] <= NEG_THRESHOLD) ? 0.f : __expf(s[nb] - m_new);
          lsum += p[nb][r];
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_row[r] = l_row[r] * corr[r] + lsum;
        m_row[r] = m_new;
      }

#pragma unroll
      for (int r = 0; r < 4; ++r)
#pragma unroll
        for (int nb = 0; nb < DKVT / 16; ++nb)
          p_lds[wave][hi * 4 + r][nb * 16 + col] = f32_to_bf16(p[nb][r]);
      __builtin_amdgcn_s_waitcnt(0);

      const bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(&p_lds[wave][col][hi * 8]);

#pragma unroll
      for (int d = 0; d < HD / 16; ++d)
#pragma unroll
        for (int r = 0; r < 4; ++r) acc_o[d][r] *= corr[r];
#pragma unroll
      for (int d = 0; d < HD / 16; ++d) {
        const bf16x8 vfrag = *reinterpret_cast<const bf16x8*>(&vt_raw[wave][DVT_BYTE(d * 16 + col, hi * 16)]);
        acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, vfrag, acc_o[d], 0, 0, 0);
      }
    }

    // ---- cross-wave combine for this head group (overlay V^T LDS with O rows)
    __syncthreads();
    float* c_o = reinterpret_cast<float*>(vt_raw[wave]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int gl = hi * 4 + r;
      if (g0 + gl < GQ) {
#pragma unroll
        for (int d = 0; d < HD / 16; ++d) c_o[gl * HD + d * 16 + col] = acc_o[d][r];
        if (col == 0) {
          c_ml[wave][gl][0] = m_row[r];
          c_ml[wave][gl][1] = l_row[r];
        }
      }
    }
    __syncthreads();

    if (wave == 0) {
      const int gn = min(16, GQ - g0);  // heads in this group
      float* po = part_o + (((size_t)bkv * n_splits + split) * GQ + g0) * HD;
      float* pml = part_ml + (((size_t)bkv * n_splits + split) * GQ + g0) * 2;
      for (int idx = lane; idx < gn * HD; idx += WAVE) {
        const int gl = idx / HD, d = idx - gl * HD;
        float m_star = c_ml[0][gl][0];
#pragma unroll
        for (int w = 1; w < DWAVES; ++w) m_star = fmaxf(m_star, c_ml[w][gl][0]);
        float osum = 0.f, lsum = 0.f;
#pragma unroll
        for (int w = 0; w < DWAVES; ++w) {
          const float wgt = (c_ml[w][gl][0] <= NEG_THRESHOLD) ? 0.f : __expf(c_ml[w][gl][0] - m_star);
          osum += wgt * reinterpret_cast<const float*>(vt_raw[w])[gl * HD + d];
          lsum += wgt * c_ml[w][gl][1];
        }
        if (out_direct) {
          // single split: normalize here (matches the combine kernel's
          // n_splits==1 output exactly) and skip that kernel entirely —
          // at short kv both kernels are pure launch latency
          out_direct[((size_t)bkv * GQ + g0 + gl) * HD + d] = osum / fmaxf(lsum, 1e-20f);
        } else {
          po[idx] = osum;
          if (d == 0) {
            pml[gl * 2 + 0] = m_star;
            pml[gl * 2 + 1] = lsum;
          }
        }
      }
    }
  }
}

template <int HD, int GQ>
__global__ __launch_bounds__(64) void attn_decode_kernel(
    const float* __restrict__ q,           // [B, KV, GQ, HD]
    const unsigned short* __restrict__ k_cache,  // [Bc, KV, Lmax, HD]
    const unsigned short* __restrict__ v_cache,
    float* __restrict__ part_o,            // [B*KV, splits, GQ, HD]
    float* __restrict__ part_ml,           // [B*KV, splits, GQ, 2]
    const int* __restrict__ kv_len_ptr,
    const float* __restrict__ alibi,  // [kv_heads*GQ] slopes or null (ALiBi adds slope*j; the
                                      // row-constant part of the bias cancels in softmax)
    int kv_heads,
    int lmax,
    int n_splits,
    float scale) {
  constexpr int LANES_PER_ROW = 16;
  constexpr int EPL = HD / LANES_PER_ROW;  // elems per lane (8 for HD=128)
  constexpr int GROUPS = WAVE / LANES_PER_ROW;  // 4 rows in flight

  const int bkv = blockIdx.x;
  const int b = bkv / kv_heads;
  const int kv = bkv - b * kv_heads;
  const int split = blockIdx.y;
  const int kv_len = *kv_len_ptr;

  const int rows_per_split = (kv_len + n_splits - 1) / n_splits;
  const int j_begin = split * rows_per_split;
  const int j_end = min(j_begin + rows_per_split, kv_len);

  const int lane = threadIdx.x & (WAVE - 1);
  const int grp = lane / LANES_PER_ROW;  // 0..3
  const int gl = lane & (LANES_PER_ROW - 1);  // lane within group

  // stage q in LDS (scaled)
  __shared__ float q_lds[GQ][HD];
  for (int idx = threadIdx.x; idx < GQ * HD; idx += WAVE) {
    q_lds[idx / HD][idx % HD] = q[((size_t)(b * kv_heads + kv) * GQ) * HD + idx] * scale;
  }
  __syncthreads();

  float m[GQ], l[GQ], o[GQ][EPL];
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
    m[g] = NEG_SENTINEL;
    l[g] = 0.f;
#pragma unroll
    for (int e = 0; e < EPL; ++e) o[g][e] = 0.f;
  }

  const unsigned short* k_base = k_cache + ((size_t)b * kv_heads + kv) * lmax * HD;
  const unsigned short* v_base = v_cache + ((size_t)b * kv_heads + kv) * lmax * HD;

  float sl[GQ];
#pragma unroll
  for (int g = 0; g < GQ; ++g) sl[g] = alibi ? alibi[kv * GQ + g] : 0.f;

  for (int j = j_begin + grp; j < j_end; j += GROUPS) {
    // load K row slice: 16 lanes x EPL elems, coalesced
    const unsigned short* krow = k_base + (size_t)j * HD + gl * EPL;
    float kf[EPL];
    if (EPL == 8) {
      const short8 k8 = *reinterpret_cast<const short8*>(krow);
#pragma unroll
      for (int e = 0; e < 8; ++e) kf[e] = bf16_to_f32((unsigned short)k8[e]);
    } else {
#pragma unroll
      for (int e = 0; e < EPL; ++e) kf[e] = bf16_to_f32(krow[e]);
    }
    // scores for each q head of this group
    float s[GQ];
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      float partial = 0.f;
#pragma unroll
      for (int e = 0; e < EPL; ++e) partial = fmaf(kf[e], q_lds[g][gl * EPL + e], partial);
      s[g] = group16_reduce_sum(partial) + sl[g] * j;  // all lanes get the row score
    }
    // V row slice
    const unsigned short* vrow = v_base + (size_t)j * HD + gl * EPL;
    float vf[EPL];
    if (EPL == 8) {
      const short8 v8 = *reinterpret_cast<const short8*>(vrow);
#pragma unroll
      for (int e = 0; e < 8; ++e) vf[e] = bf16_to_f32((unsigned short)v8[e]);
    } else {
#pragma unroll
      for (int e = 0; e < EPL; ++e) vf[e] = bf16_to_f32(vrow[e]);
    }
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      const float m_new = fmaxf(m[g], s[g]);
      const float corr = __expf(m[g] - m_new);
      const float p = __expf(s[g] - m_new);
      l[g] = l[g] * corr + p;
#pragma unroll
      for (int e = 0; e < EPL; ++e) o[g][e] = o[g][e] * corr + p * vf[e];
      m[g] = m_new;
    }
  }

  // combine the 4 row-groups through LDS (deterministic order)
  __shared__ float c_m[GROUPS][GQ];
  __shared__ float c_l[GROUPS][GQ];
  __shared__ float c_o[GROUPS][GQ][HD];
  if (gl * EPL < HD) {
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
#pragma unroll
      for (int e = 0; e < EPL; ++e) c_o[grp][g][gl * EPL + e] = o[g][e];
      if (gl == 0) {
        c_m[grp][g] = m[g];
        c_l[grp][g] = l[g];
      }
    }
  }
  __syncthreads();

  float* po = part_o + (((size_t)bkv * n_splits + split) * GQ) * HD;
  float* pml = part_ml + (((size_t)bkv * n_splits + split) * GQ) * 2;
  for (int idx = threadIdx.x; idx < GQ * HD; idx += WAVE) {
    const int g = idx / HD, d = idx % HD;
    float m_star = c_m[0][g];
#pragma unroll
    for (int r = 1; r < GROUPS; ++r) m_star = fmaxf(m_star, c_m[r][g]);
    float osum = 0.f, lsum = 0.f;
#pragma unroll
    for (int r = 0; r < GROUPS; ++r) {
      const float w = (c_m[r][g] <= NEG_THRESHOLD) ? 0.f : __expf(c_m[r][g] - m_star);
      osum += w * c_o[r][g][d];
      lsum += w * c_l[r][g];
    }
    po[idx] = osum;
    if (d == 0) {
      pml[g * 2 + 0] = m_star;
      pml[g * 2 + 1] = lsum;
    }
  }
}

__global__ void attn_decode_combine_kernel(
    const float* __restrict__ part_o,   // [B*KV, splits, GQ, HD]
    const float* __restrict__ part_ml,  // [B*KV, splits, GQ, 2]
    float* __restrict__ out,            // [B, KV*GQ*HD]
    int n_splits, int gq, int hd) {
  const int bkv = blockIdx.x;
  const int g = blockIdx.y;
  float m_star = NEG_SENTINEL;
  for (int s = 0; s < n_splits; ++s)
    m_star = fmaxf(m_star, part_ml[(((size_t)bkv * n_splits + s) * gq + g) * 2]);
  for (int d = threadIdx.x; d < hd; d += blockDim.x) {
    float osum = 0.f, lsum = 0.f;
    for (int s = 0; s < n_splits; ++s) {
      const float ms = part_ml[(((size_t)bkv * n_splits + s) * gq + g) * 2];
      const float ls = part_ml[(((size_t)bkv * n_splits + s) * gq + g) * 2 + 1];
      if (ms <= NEG_THRESHOLD) continue;
      const float w = __expf(ms - m_star);
      osum += w * part_o[(((size_t)bkv * n_splits + s) * gq + g) * hd + d];
      lsum += w * ls;
    }
    out[(size_t)bkv * gq * hd + (size_t)g * hd + d] = osum / fmaxf(lsum, 1e-20f);
  }
}

// ---------------------------------------------------------------- host API

// q: [B, KV*GQ*HD] f32 (token per row); caches [Bc, KV, Lmax, HD] bf16;
// kv_len: device int32 scalar tensor. Returns [B, KV*GQ*HD] f32.
torch::Tensor attn_decode_fused(
    torch::Tensor q,
    torch::Tensor k_cache,
    torch::Tensor v_cache,
    torch::Tensor kv_len,  // device int32 [1]
    int64_t gq,
    int64_t n_splits_i,
    torch::Tensor part_o,   // workspace [B*KV, splits, GQ, HD] f32 (or empty)
    torch::Tensor part_ml,  // workspace [B*KV, splits, GQ, 2] f32 (or empty)
    double scale,
    c10::optional<torch::Tensor> alibi_slopes) {  // [kv_heads*GQ] f32
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kFloat32);
  TORCH_CHECK(k_cache.dtype() == torch::kBFloat16 && k_cache.dim() == 4);
  const int b = q.size(0);
  const int kv_heads = k_cache.size(1), lmax = k_cache.size(2), hd = k_cache.size(3);
  const int GQi = (int)gq;
  static const bool use_valu = [] {
    const char* s = std::getenv("PETALS_DECODE_KERNEL");
    return s && s[0] == 'v';
  }();
  int n_splits = (int)n_splits_i;
  if (n_splits <= 0) {
    const int bkv = std::max(1, b * kv_heads);
    if (use_valu) {
      // VALU kernel: target ~1024 single-wave workgroups, chunks >= 64 rows
      n_splits = std::max(1, std::min(1024 / bkv, (lmax + 63) / 64));
    } else {
      // MFMA kernel: 4-wave workgroups; fill ~256 CUs (chunks >= 128 rows) and
      // go up to 3 wgs/CU only when chunks stay >= 512 rows (measured knees,
      // profiles/decode_longctx_sweep.log + scripts/decode_mfma.hip)
      const int by_occ = std::min(768 / bkv, lmax / 512);
      const int by_min = std::min(256 / bkv, (lmax + 127) / 128);
      n_splits = std::max(1, std::max(by_occ, by_min));
      // measured dead end (profiles/attn_direct_ab.log): forcing one split
      // + in-kernel normalization (combine kernel skipped) at short caches
      // LOST 1.4 tok/s end to end — halving the split occupancy costs more
      // than the removed launch. Off by default; the env knob keeps the
      // single-split direct-out path selectable for re-evaluation.
      static const int short_max = [] {
        const char* e = std::getenv("PETALS_AMD_DECODE_SHORT_LMAX");
        return e ? std::atoi(e) : 0;
      }();
      if (lmax <= short_max) n_splits = 1;
    }
  }
  auto opts = q.options();
  const bool will_direct = !use_valu && n_splits == 1;  // parts go unused
  if (!will_direct && part_o.numel() < (int64_t)b * kv_heads * n_splits * GQi * hd)
    part_o = torch::empty({(int64_t)b * kv_heads, n_splits, GQi, hd}, opts);
  if (!will_direct && part_ml.numel() < (int64_t)b * kv_heads * n_splits * GQi * 2)
    part_ml = torch::empty({(int64_t)b * kv_heads, n_splits, GQi, 2}, opts);
  auto out = torch::empty({(int64_t)b, (int64_t)kv_heads * GQi * hd}, opts);

  dim3 grid(b * kv_heads, n_splits);
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* qp = q.data_ptr<float>();
  const unsigned short* kp = reinterpret_cast<const unsigned short*>(k_cache.data_ptr());
  const unsigned short* vp = reinterpret_cast<const unsigned short*>(v_cache.data_ptr());
  float* pop = part_o.data_ptr<float>();
  float* pmlp = part_ml.data_ptr<float>();
  const int* lenp = kv_len.data_ptr<int>();
  const float sc = (float)scale;
  const float* alibi_p = nullptr;
  if (alibi_slopes.has_value() && alibi_slopes->defined() && alibi_slopes->numel() > 0) {
    TORCH_CHECK(alibi_slopes->is_contiguous() && alibi_slopes->dtype() == torch::kFloat32);
    TORCH_CHECK(alibi_slopes->numel() == (int64_t)kv_heads * GQi, "alibi slopes must be [q_heads]");
    alibi_p = alibi_slopes->data_ptr<float>();
  }

  // single split on the MFMA path: the decode kernel normalizes and writes
  // the output itself — the combine kernel (pure launch latency at short kv)
  // is skipped entirely
  float* direct_p = (!use_valu && n_splits == 1) ? out.data_ptr<float>() : nullptr;
  bool launched = false;
#define ATTN_CASE(HDV, GQV)                                                   \
  if (hd == HDV && GQi == GQV) {                                              \
    if (use_valu)                                                             \
      attn_decode_kernel<HDV, GQV><<<grid, 64, 0, stream>>>(                  \
          qp, kp, vp, pop, pmlp, lenp, alibi_p, kv_heads, lmax, n_splits, sc);\
    else                                                                      \
      mfma_decode_kernel<HDV, GQV><<<grid, DWAVES * WAVE, 0, stream>>>(       \
          qp, kp, vp, pop, pmlp, lenp, alibi_p, kv_heads, lmax, n_splits, sc, \
          direct_p);                                                          \
    launched = true;                                                          \
  }
  ATTN_CASE(128, 1) ATTN_CASE(128, 2) ATTN_CASE(128, 4) ATTN_CASE(128, 6)
  ATTN_CASE(128, 8) ATTN_CASE(128, 16)
  ATTN_CASE(64, 1) ATTN_CASE(64, 2) ATTN_CASE(64, 4) ATTN_CASE(64, 8) ATTN_CASE(64, 16)
#undef ATTN_CASE
  // GQ > 16 (multi-group MFMA path only; the VALU kernel would blow its
  // per-lane register arrays): falcon-7b MQA (gq=71), falcon-180b (gq=29)
#define ATTN_CASE_BIG(HDV, GQV)                                               \
  if (hd == HDV && GQi == GQV) {                                              \
    TORCH_CHECK(!use_valu, "VALU decode kernel does not support gq=", GQV);   \
    mfma_decode_kernel<HDV, GQV><<<grid, DWAVES * WAVE, 0, stream>>>(         \
        qp, kp, vp, pop, pmlp, lenp, alibi_p, kv_heads, lmax, n_splits, sc,   \
        direct_p);                                                            \
    launched = true;                                                          \
  }
  ATTN_CASE_BIG(64, 29) ATTN_CASE_BIG(64, 71) ATTN_CASE_BIG(64, 32)
  ATTN_CASE_BIG(128, 32)
#undef ATTN_CASE_BIG
  TORCH_CHECK(launched, "unsupported (head_dim, gqa) = (", hd, ", ", GQi, ")");
  HIP_CHECK_LAST();

  if (direct_p == nullptr) {
    dim3 cgrid(b * kv_heads, GQi);
    attn_decode_combine_kernel<<<cgrid, std::min(hd, 256), 0, stream>>>(
        pop, pmlp, out.data_ptr<float>(), n_splits, GQi, hd);
    HIP_CHECK_LAST();
  }
  return out;
}
