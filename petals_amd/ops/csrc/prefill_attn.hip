// Flash-style MFMA prefill attention for CDNA4 (gfx950), bf16, causal, GQA.
//
// Geometry: one workgroup = 4 waves = 64 query rows of ONE (batch, q_head);
// each wave owns a 16-row q tile held in registers as mfma_f32_16x16x32_bf16
// A-fragments. K/V tiles of 32 keys are staged in LDS by the whole workgroup
// (K row-major padded; V transposed so the PV B-fragments are contiguous
// ds_read_b128 rows). Per tile: 8 QK^T MFMAs -> causal mask -> online softmax
// (per-lane row stats: the C-layout keeps each q row's 4 accumulator rows in
// the same lane) -> P through a per-wave LDS round-trip into A-fragment
// layout -> 8 PV MFMAs. Fragment layouts HW-verified by scripts/mfma_verify.hip.
//
// Replaces the chunked rocBLAS-matmul + fp32-softmax prefill composition
// (ops/reference.py attention) on the GPU path; the reference framework used
// torch/CUDA attention inside HF blocks (reference models/llama/block.py:108).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QTILE 16   // q rows per wave
#define WAVES 4    // waves per workgroup
#define KVTILE 32  // keys per LDS tile
#define KPAD 8     // LDS row padding (elements) against bank conflicts

template <int HD>
__global__ __launch_bounds__(WAVES * 64) void attn_prefill_kernel(
    const unsigned short* __restrict__ q,   // [B, QH, S, HD]
    const unsigned short* __restrict__ k,   // [B, KVH, Lmax, HD]
    const unsigned short* __restrict__ v,   // [B, KVH, Lmax, HD]
    unsigned short* __restrict__ out,       // [B, QH, S, HD]
    int q_heads,
    int kv_heads,
    int s_q,
    int lmax,
    int kv_len,     // keys to attend over (cache valid prefix)
    int kv_offset,  // absolute position of q row 0 within the kv sequence
    float scale,
    int causal) {
  const int bh = blockIdx.x;          // b * q_heads + qh
  const int b = bh / q_heads;
  const int qh = bh - b * q_heads;
  const int kvh = qh / (q_heads / kv_heads);
  const int q0_wg = blockIdx.y * (WAVES * QTILE);  // first q row of this wg

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;      // fragment column (and C col)
  const int hi = lane >> 4;       // fragment k-group (and C row group)
  const int q0 = q0_wg + wave * QTILE;  // this wave's first q row

  // LDS: K tile row-major [KVTILE][HD+KPAD]; V tile transposed [HD][KVTILE+KPAD]
  // with an XOR swizzle on the dim-group (rows 8 apart land 0 mod 128 B with a
  // 16 B-aligned stride, so the scalar transpose writes were 16-way
  // bank-conflicted — the dominant cost in the v1 PMC profile); per-wave P
  // scratch [QTILE][KVTILE+KPAD]
  __shared__ unsigned short k_lds[KVTILE][HD + KPAD];
  __shared__ unsigned char vt_raw[HD * (KVTILE + KPAD) * 2];
  __shared__ unsigned short p_lds[WAVES][QTILE][KVTILE + KPAD];
#define VT_BYTE(dim, key_byte)   (((unsigned)(dim) * ((KVTILE + KPAD) * 2) + (unsigned)(key_byte)) ^ ((((unsigned)(dim) >> 3) & 7u) << 4))

  constexpr int KCH = HD / 32;  // 32-wide k-dim chunks per head dim

  // ---- load this wave's q tile into A-fragments (zero-padded past s_q)
  bf16x8 q_frag[KCH];
  const size_t q_base = (((size_t)b * q_heads + qh) * s_q) * HD;
  const int my_qrow = q0 + col;  // A: row = lane&15
#pragma unroll
  for (int kc = 0; kc < KCH; ++kc) {
    if (my_qrow < s_q) {
      const unsigned short* src = q + q_base + (size_t)my_qrow * HD + kc * 32 + hi * 8;
      q_frag[kc] = *reinterpret_cast<const bf16x8*>(src);
    } else {
      q_frag[kc] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // ---- accumulators: O in C-layout (8 dim-blocks x f32x4), softmax stats
  f32x4 acc_o[HD / 16];
#pragma unroll
  for (int d = 0; d < HD / 16; ++d) acc_o[d] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_row[4], l_row[4];  // for rows hi*4 + r (C layout rows of this lane)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_row[r] = NEG_SENTINEL;
    l_row[r] = 0.f;
  }

  // causal bound for this WORKGROUP (max key any of its q rows may see)
  const int wg_last_q_abs = kv_offset + min(q0_wg + WAVES * QTILE, s_q) - 1;
  const int kv_end = causal ? min(kv_len, wg_last_q_abs + 1) : kv_len;

  const size_t kv_base = (((size_t)b * kv_heads + kvh) * lmax) * HD;

  for (int j0 = 0; j0 < kv_end; j0 += KVTILE) {
    const int tile_n = min(KVTILE, kv_end - j0);
    // ---- stage K tile (row-major) and V tile (transposed), 256 threads
    __syncthreads();
    for (int idx = tid; idx < KVTILE * (HD / 8); idx += WAVES * WAVE) {
      const int row = idx / (HD / 8);
      const int c8 = (idx - row * (HD / 8)) * 8;
      bf16x8 kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (j0 + row < kv_end) {
        kv8 = *reinterpret_cast<const bf16x8*>(k + kv_base + (size_t)(j0 + row) * HD + c8);
        vv8 = *reinterpret_cast<const bf16x8*>(v + kv_base + (size_t)(j0 + row) * HD + c8);
      }
      *reinterpret_cast<bf16x8*>(&k_lds[row][c8]) = kv8;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        *reinterpret_cast<unsigned short*>(&vt_raw[VT_BYTE(c8 + e, row * 2)]) = (unsigned short)vv8[e];
    }
    __syncthreads();

    // ---- S = Q K^T : two 16-key column blocks
    f32x4 s_acc[2];
#pragma unroll
    for (int nb = 0; nb < 2; ++nb) {
      s_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < KCH; ++kc) {
        // B[k][n] = K[key = nb*16 + col][kdim = kc*32 + hi*8 + reg]
        const bf16x8 kt = *reinterpret_cast<const bf16x8*>(&k_lds[nb * 16 + col][kc * 32 + hi * 8]);
        s_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[kc], kt, s_acc[nb], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax (per-lane rows hi*4+r, col = key)
    float p[2][4];  // [nb][r] probabilities for this lane's slots
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + hi * 4 + r;
      const int q_abs = kv_offset + qrow;
      float s0 = s_acc[0][r] * scale;
      float s1 = s_acc[1][r] * scale;
      const int key0 = j0 + col, key1 = j0 + 16 + col;
      const bool dead0 = key0 >= tile_n + j0 || (causal && key0 > q_abs) || qrow >= s_q;
      const bool dead1 = key1 >= tile_n + j0 || (causal && key1 > q_abs) || qrow >= s_q;
      if (dead0) s0 = NEG_SENTINEL;
      if (dead1) s1 = NEG_SENTINEL;
      // row max across the 16 lanes holding this row (xor within low 4 bits)
      float mx = fmaxf(s0, s1);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      const float m_new = fmaxf(m_row[r], mx);
      corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
      const float p0 = (s0 <= NEG_THRESHOLD) ? 0.f : __expf(s0 - m_new);
      const float p1 = (s1 <= NEG_THRESHOLD) ? 0.f : __expf(s1 - m_new);
      float lsum = p0 + p1;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
      l_row[r] = l_row[r] * corr[r] + lsum;
      m_row[r] = m_new;
      p[0][r] = p0;
      p[1][r] = p1;
    }

    // ---- write P (bf16) into per-wave LDS in [q_row][key] layout
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      p_lds[wave][hi * 4 + r][col] = f32_to_bf16(p[0][r]);
      p_lds[wave][hi * 4 + r][16 + col] = f32_to_bf16(p[1][r]);
    }
    // the P round-trip is per-wave, but a block barrier is the simple safe
    // ordering (the compiler may not prove the write/read regions disjoint)
    __syncthreads();

    // P A-fragment: A[row = lane&15][k = hi*8 + reg] over the 32 keys
    const bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(&p_lds[wave][col][hi * 8]);

    // ---- rescale O, then PV
#pragma unroll
    for (int d = 0; d < HD / 16; ++d)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[d][r] *= corr[r];
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) {
      // B[k = key][n = dim] = VT[dim = d*16 + col][key = hi*8 + reg]
      const bf16x8 vfrag =
          *reinterpret_cast<const bf16x8*>(&vt_raw[VT_BYTE(d * 16 + col, hi * 16)]);
      acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, vfrag, acc_o[d], 0, 0, 0);
    }
  }

  // ---- write O: C layout row = hi*4 + r, col; scale by 1/l
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + hi * 4 + r;
    if (qrow >= s_q) continue;
    const float inv_l = l_row[r] > 0.f ? 1.0f / l_row[r] : 0.f;
    unsigned short* dst = out + q_base + (size_t)qrow * HD;
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) dst[d * 16 + col] = f32_to_bf16(acc_o[d][r] * inv_l);
  }
}

torch::Tensor attn_prefill_fused(
    torch::Tensor q,  // [B, QH, S, HD] bf16
    torch::Tensor k,  // [B, KVH, Lmax, HD] bf16 (cache; valid prefix kv_len)
    torch::Tensor v,
    int64_t kv_len,
    int64_t kv_offset,
    double scale,
    bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.dim() == 4);
  TORCH_CHECK(k.dtype() == torch::kBFloat16 && k.dim() == 4);
  const int B = q.size(0), QH = q.size(1), S = q.size(2), HD = q.size(3);
  const int KVH = k.size(1), LMAX = k.size(2);
  TORCH_CHECK(HD == 128 || HD == 64, "prefill attention supports head_dim 64/128");
  TORCH_CHECK(QH % KVH == 0);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  auto out = torch::empty_like(q);
  dim3 grid(B * QH, (S + WAVES * QTILE - 1) / (WAVES * QTILE));
  auto stream = at::cuda::getCurrentCUDAStream();
  const float sc = (float)scale;

  if (HD == 128) {
    attn_prefill_kernel<128><<<grid, WAVES * WAVE, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(q.data_ptr()),
        reinterpret_cast<const unsigned short*>(k.data_ptr()),
        reinterpret_cast<const unsigned short*>(v.data_ptr()),
        reinterpret_cast<unsigned short*>(out.data_ptr()),
        QH, KVH, S, LMAX, (int)kv_len, (int)kv_offset, sc, causal ? 1 : 0);
  } else {
    attn_prefill_kernel<64><<<grid, WAVES * WAVE, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(q.data_ptr()),
        reinterpret_cast<const unsigned short*>(k.data_ptr()),
        reinterpret_cast<const unsigned short*>(v.data_ptr()),
        reinterpret_cast<unsigned short*>(out.data_ptr()),
        QH, KVH, S, LMAX, (int)kv_len, (int)kv_offset, sc, causal ? 1 : 0);
  }
  HIP_CHECK_LAST();
  return out;
}
