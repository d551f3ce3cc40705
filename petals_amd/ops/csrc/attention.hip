// GQA decode attention for CDNA4 (gfx950), flash-decoding style:
// grid = (batch x kv_heads, kv splits); each 64-thread block streams a slice
// of the KV cache once (coalesced 16 B/lane K/V row reads by 16-lane groups),
// keeps online-softmax state per q-head in registers, block-combines its 4
// row-groups through LDS, and writes one unnormalized partial per split.
// attn_decode_combine merges the splits. kv_len comes from a DEVICE pointer so
// the whole decode step can be captured in a hipGraph with a moving position.
//
// Replaces the reference's torch QK^T/softmax/AV decode math
// (reference models/llama/block.py:108-127) on the MI355X fast path.

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

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
  int n_splits = (int)n_splits_i;
  if (n_splits <= 0) {
    // target ~1024 single-wave workgroups; each split should keep >=64 cache
    // rows so the streaming loop amortizes setup
    n_splits = std::max(1, std::min(1024 / std::max(1, b * kv_heads), (lmax + 63) / 64));
  }
  auto opts = q.options();
  if (part_o.numel() < (int64_t)b * kv_heads * n_splits * GQi * hd)
    part_o = torch::empty({(int64_t)b * kv_heads, n_splits, GQi, hd}, opts);
  if (part_ml.numel() < (int64_t)b * kv_heads * n_splits * GQi * 2)
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

  bool launched = false;
#define ATTN_CASE(HDV, GQV)                                                   \
  if (hd == HDV && GQi == GQV) {                                              \
    attn_decode_kernel<HDV, GQV><<<grid, 64, 0, stream>>>(                    \
        qp, kp, vp, pop, pmlp, lenp, alibi_p, kv_heads, lmax, n_splits, sc);  \
    launched = true;                                                          \
  }
  ATTN_CASE(128, 1) ATTN_CASE(128, 2) ATTN_CASE(128, 4) ATTN_CASE(128, 6)
  ATTN_CASE(128, 8) ATTN_CASE(128, 16)
  ATTN_CASE(64, 1) ATTN_CASE(64, 2) ATTN_CASE(64, 4) ATTN_CASE(64, 8) ATTN_CASE(64, 16)
#undef ATTN_CASE
  TORCH_CHECK(launched, "unsupported (head_dim, gqa) = (", hd, ", ", GQi, ")");
  HIP_CHECK_LAST();

  dim3 cgrid(b * kv_heads, GQi);
  attn_decode_combine_kernel<<<cgrid, std::min(hd, 256), 0, stream>>>(
      pop, pmlp, out.data_ptr<float>(), n_splits, GQi, hd);
  HIP_CHECK_LAST();
  return out;
}
