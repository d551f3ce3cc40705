// Flash-style MFMA prefill attention for CDNA4 (gfx950), bf16, causal, GQA.
//
// Geometry: one workgroup = 4 waves = 64 query rows of ONE (batch, q_head);
// each wave owns a 16-row q tile held in registers as mfma_f32_16x16x32_bf16
// A-fragments. K/V tiles of KVT keys are staged in LDS by the whole workgroup
// (K row-major padded; V transposed so the PV B-fragments are contiguous
// ds_read_b128 rows). Per tile: KVT/16*KCH QK^T MFMAs -> causal mask -> online
// softmax (per-lane row stats: the C-layout keeps each q row's 4 accumulator
// rows in the same lane) -> P through a per-wave LDS round-trip into
// A-fragment layout -> KVT/32*HD/16 PV MFMAs. Fragment layouts HW-verified by
// scripts/mfma_verify.hip. KVT=64 default (amortizes softmax shuffles and
// barriers over 2x the MFMA work vs KVT=32); PETALS_PREFILL_KVT=32 selects
// the narrow tile for A/B.
//
// Replaces the chunked rocBLAS-matmul + fp32-softmax prefill composition
// (ops/reference.py attention) on the GPU path; the reference framework used
// torch/CUDA attention inside HF blocks (reference models/llama/block.py:108).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdlib>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QTILE 16  // q rows per wave
#define WAVES 4   // waves per workgroup
#define KPAD 8    // LDS row padding (elements) against bank conflicts

// V-transpose LDS addressing: XOR swizzle on the dim-group (rows 8 apart land
// 0 mod 128 B with any 16 B-aligned pitch, so the scalar transpose writes were
// 16-way bank-conflicted — the dominant cost in the v1 PMC profile). Verified
// injective with stage-write multiplicity 2 (b16 ideal) and PV-read
// multiplicity 8 (b128 ideal) for both KVT=32 and KVT=64.
#define VT_BYTE(dim, key_byte) \
  ((((unsigned)(dim)) * ((KVT + KPAD) * 2) + (unsigned)(key_byte)) ^ ((((unsigned)(dim) >> 3) & 7u) << 4))

// QR = q rows per wave (16 or 32). QR=32 runs NSET=2 16-row MFMA row sets per
// wave off ONE K/V staging pass: staging traffic and barriers per flop halve,
// at the cost of ~2x accumulator VGPRs and a larger per-wave P scratch.
template <int HD, int KVT, bool ALIBI, int QR>
__global__ __launch_bounds__(WAVES * 64) void attn_prefill_kernel(
    const unsigned short* __restrict__ q,   // [B, QH, S, HD]
    const unsigned short* __restrict__ k,   // [B, KVH, Lmax, HD]
    const unsigned short* __restrict__ v,   // [B, KVH, Lmax, HD]
    unsigned short* __restrict__ out,       // [B, QH, S, HD]
    const float* __restrict__ alibi,        // [QH] ALiBi slopes or null (adds slope*key_abs;
                                            // the per-row constant cancels in softmax)
    int q_heads,
    int kv_heads,
    int s_q,
    int lmax,
    int kv_len,     // keys to attend over (cache valid prefix)
    int kv_offset,  // absolute position of q row 0 within the kv sequence
    float scale,
    int causal) {
  constexpr int NSET = QR / 16;  // 16-row MFMA row sets per wave
  const int bh = blockIdx.x;          // b * q_heads + qh
  const int b = bh / q_heads;
  const int qh = bh - b * q_heads;
  const int kvh = qh / (q_heads / kv_heads);
  const int q0_wg = blockIdx.y * (WAVES * QR);  // first q row of this wg

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;      // fragment column (and C col)
  const int hi = lane >> 4;       // fragment k-group (and C row group)
  const int q0 = q0_wg + wave * QR;  // this wave's first q row
  // softmax runs in the log2 domain (exp2 instead of exp): q is pre-scaled by
  // scale*log2e at load, saving a VALU mul per score AND per exponential —
  // the kernel was issue-bound on VALU at 17:1 VALU:MFMA (PMC, round 1)
  const float slope = ALIBI ? alibi[qh] * 1.4426950408889634f : 0.f;  // wave-uniform

  // LDS: K tile row-major [KVT][HD+KPAD]; V tile transposed+swizzled
  // [HD][KVT+KPAD]; per-wave P scratch [QR][KVT+KPAD]
  __shared__ unsigned short k_lds[KVT][HD + KPAD];
  __shared__ unsigned char vt_raw[HD * (KVT + KPAD) * 2];
  __shared__ unsigned short p_lds[WAVES][QR][KVT + KPAD];

  constexpr int KCH = HD / 32;   // 32-wide k-dim chunks per head dim
  constexpr int NB = KVT / 16;   // 16-key S column blocks per tile
  constexpr int PKC = KVT / 32;  // 32-key PV k-chunks per tile

  // ---- load this wave's q tiles into A-fragments (zero-padded past s_q),
  // pre-scaled by scale*log2e (see the log2-domain softmax note above)
  const float qscale = scale * 1.4426950408889634f;
  bf16x8 q_frag[NSET][KCH];
  const size_t q_base = (((size_t)b * q_heads + qh) * s_q) * HD;
#pragma unroll
  for (int qs = 0; qs < NSET; ++qs) {
    const int my_qrow = q0 + qs * 16 + col;  // A: row = lane&15
#pragma unroll
    for (int kc = 0; kc < KCH; ++kc) {
      if (my_qrow < s_q) {
        const unsigned short* src = q + q_base + (size_t)my_qrow * HD + kc * 32 + hi * 8;
        const bf16x8 raw = *reinterpret_cast<const bf16x8*>(src);
        short vs[8];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          vs[e] = (short)f32_to_bf16(bf16_to_f32((unsigned short)raw[e]) * qscale);
        q_frag[qs][kc] = bf16x8{vs[0], vs[1], vs[2], vs[3], vs[4], vs[5], vs[6], vs[7]};
      } else {
        q_frag[qs][kc] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  // ---- accumulators: O in C-layout (HD/16 dim-blocks x f32x4), softmax stats
  f32x4 acc_o[NSET][HD / 16];
  float m_row[NSET][4], l_row[NSET][4];  // rows qs*16 + hi*4 + r (C layout)
#pragma unroll
  for (int qs = 0; qs < NSET; ++qs) {
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) acc_o[qs][d] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_row[qs][r] = NEG_SENTINEL;
      l_row[qs][r] = 0.f;
    }
  }

  // causal bound for this WORKGROUP (max key any of its q rows may see)
  const int wg_last_q_abs = kv_offset + min(q0_wg + WAVES * QR, s_q) - 1;
  const int kv_end = causal ? min(kv_len, wg_last_q_abs + 1) : kv_len;

  const size_t kv_base = (((size_t)b * kv_heads + kvh) * lmax) * HD;

  for (int j0 = 0; j0 < kv_end; j0 += KVT) {
    const int tile_n = min(KVT, kv_end - j0);
    // ---- stage K tile (row-major) and V tile (transposed), 256 threads
    __syncthreads();
    for (int idx = tid; idx < KVT * (HD / 8); idx += WAVES * WAVE) {
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

    float corr[NSET][4];
#pragma unroll
    for (int qs = 0; qs < NSET; ++qs) {
      const int q0s = q0 + qs * 16;  // first q row of this 16-row set
      // ---- S = Q K^T : NB 16-key column blocks
      f32x4 s_acc[NB];
#pragma unroll
      for (int nb = 0; nb < NB; ++nb) {
        s_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kc = 0; kc < KCH; ++kc) {
          // B[k][n] = K[key = nb*16 + col][kdim = kc*32 + hi*8 + reg]
          const bf16x8 kt = *reinterpret_cast<const bf16x8*>(&k_lds[nb * 16 + col][kc * 32 + hi * 8]);
          s_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[qs][kc], kt, s_acc[nb], 0, 0, 0);
        }
      }

      // ---- causal mask + online softmax (per-lane rows hi*4+r, col = key)
      float p[NB][4];  // [nb][r] probabilities for this lane's slots
      // interior tiles (no tail, no causal edge, no s_q edge for ANY row of
      // this set) skip all masking VALU: +10% at S=4096 (PMC showed the
      // kernel issue-bound on VALU at 17:1 VALU:MFMA)
      const bool interior =
          (tile_n == KVT) && (q0s + QTILE <= s_q) && (!causal || j0 + KVT <= kv_offset + q0s + 1);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0s + hi * 4 + r;
        const int q_abs = kv_offset + qrow;
        // NOTE: masking and the running max are deliberately SEPARATE loops.
        // Fusing them (mask + `mx = fmaxf(mx, s[nb])` in one loop body) makes
        // amdclang (ROCm 7.2, gfx950, -O3, with or without -ffast-math)
        // miscompile the conditional sentinel store: dead keys keep their raw
        // scores for the r==0 slot of every accumulator row group, leaking
        // masked keys into the softmax (bisected in scripts/prefill_bisect.hip).
        float s[NB];
        if (interior) {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) {
            s[nb] = s_acc[nb][r];  // already scaled (q pre-scaled by scale*log2e)
            if (ALIBI) s[nb] += slope * (j0 + nb * 16 + col);
          }
        } else {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) {
            const int key = j0 + nb * 16 + col;
            s[nb] = s_acc[nb][r];
            if (ALIBI) s[nb] += slope * key;
            if (key >= tile_n + j0 || (causal && key > q_abs) || qrow >= s_q) s[nb] = NEG_SENTINEL;
          }
        }
        float mx = NEG_SENTINEL;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) mx = fmaxf(mx, s[nb]);
        // row max across the 16 lanes holding this row (xor within low 4 bits)
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_row[qs][r], mx);
        corr[qs][r] = (m_row[qs][r] <= NEG_THRESHOLD) ? 0.f : exp2f(m_row[qs][r] - m_new);
        float lsum = 0.f;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          p[nb][r] = (s[nb] <= NEG_THRESHOLD) ? 0.f : exp2f(s[nb] - m_new);
          lsum += p[nb][r];
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_row[qs][r] = l_row[qs][r] * corr[qs][r] + lsum;
        m_row[qs][r] = m_new;
      }

      // ---- write P (bf16) into per-wave LDS in [q_row][key] layout
#pragma unroll
      for (int r = 0; r < 4; ++r)
#pragma unroll
        for (int nb = 0; nb < NB; ++nb)
          p_lds[wave][qs * 16 + hi * 4 + r][nb * 16 + col] = f32_to_bf16(p[nb][r]);
    }
    // the P round-trip is per-wave, but a block barrier is the simple safe
    // ordering (the compiler may not prove the write/read regions disjoint);
    // ONE barrier covers all NSET row sets
    __syncthreads();

#pragma unroll
    for (int qs = 0; qs < NSET; ++qs) {
      // P A-fragments: A[row = lane&15][k = pk*32 + hi*8 + reg] over KVT keys
      bf16x8 p_frag[PKC];
#pragma unroll
      for (int pk = 0; pk < PKC; ++pk)
        p_frag[pk] = *reinterpret_cast<const bf16x8*>(&p_lds[wave][qs * 16 + col][pk * 32 + hi * 8]);

      // ---- rescale O, then PV
#pragma unroll
      for (int d = 0; d < HD / 16; ++d)
#pragma unroll
        for (int r = 0; r < 4; ++r) acc_o[qs][d][r] *= corr[qs][r];
#pragma unroll
      for (int d = 0; d < HD / 16; ++d) {
#pragma unroll
        for (int pk = 0; pk < PKC; ++pk) {
          // B[k = key][n = dim] = VT[dim = d*16 + col][key = pk*32 + hi*8 + reg]
          const bf16x8 vfrag =
              *reinterpret_cast<const bf16x8*>(&vt_raw[VT_BYTE(d * 16 + col, pk * 64 + hi * 16)]);
          acc_o[qs][d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag[pk], vfrag, acc_o[qs][d], 0, 0, 0);
        }
      }
    }
  }

  // ---- write O: C layout row = qs*16 + hi*4 + r, col; scale by 1/l
#pragma unroll
  for (int qs = 0; qs < NSET; ++qs)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + qs * 16 + hi * 4 + r;
      if (qrow >= s_q) continue;
      const float inv_l = l_row[qs][r] > 0.f ? 1.0f / l_row[qs][r] : 0.f;
      unsigned short* dst = out + q_base + (size_t)qrow * HD;
#pragma unroll
      for (int d = 0; d < HD / 16; ++d) dst[d * 16 + col] = f32_to_bf16(acc_o[qs][d][r] * inv_l);
    }
}

torch::Tensor attn_prefill_fused(
    torch::Tensor q,  // [B, QH, S, HD] bf16
    torch::Tensor k,  // [B, KVH, Lmax, HD] bf16 (cache; valid prefix kv_len)
    torch::Tensor v,
    int64_t kv_len,
    int64_t kv_offset,
    double scale,
    bool causal,
    c10::optional<torch::Tensor> alibi_slopes) {  // [QH] f32
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.dim() == 4);
  TORCH_CHECK(k.dtype() == torch::kBFloat16 && k.dim() == 4);
  const int B = q.size(0), QH = q.size(1), S = q.size(2), HD = q.size(3);
  const int KVH = k.size(1), LMAX = k.size(2);
  TORCH_CHECK(HD == 128 || HD == 64, "prefill attention supports head_dim 64/128");
  TORCH_CHECK(QH % KVH == 0);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  auto out = torch::empty_like(q);
  auto stream = at::cuda::getCurrentCUDAStream();
  const float sc = (float)scale;

  static const int kvt_env = [] {
    const char* s = std::getenv("PETALS_PREFILL_KVT");
    return s ? std::atoi(s) : 64;
  }();
  static const int qr_env = [] {
    const char* s = std::getenv("PETALS_PREFILL_QROWS");
    return s ? std::atoi(s) : 16;
  }();
  // QR=32 halves K/V staging + barriers per flop but needs >=8 row-tiles to
  // fill the chip for short sequences; use it only when the grid stays large
  const int qr = (qr_env == 32 && S >= 2 * WAVES * 32) ? 32 : 16;
  dim3 grid(B * QH, (S + WAVES * qr - 1) / (WAVES * qr));

  const float* alibi_p = nullptr;
  if (alibi_slopes.has_value() && alibi_slopes->defined() && alibi_slopes->numel() > 0) {
    TORCH_CHECK(alibi_slopes->is_contiguous() && alibi_slopes->dtype() == torch::kFloat32);
    TORCH_CHECK(alibi_slopes->numel() == QH, "alibi slopes must be [q_heads]");
    alibi_p = alibi_slopes->data_ptr<float>();
  }
  const auto* qp = reinterpret_cast<const unsigned short*>(q.data_ptr());
  const auto* kp = reinterpret_cast<const unsigned short*>(k.data_ptr());
  const auto* vp = reinterpret_cast<const unsigned short*>(v.data_ptr());
  auto* op = reinterpret_cast<unsigned short*>(out.data_ptr());

#define PF_LAUNCH(HDV, KVTV, AB, QRV)                                                       \
  attn_prefill_kernel<HDV, KVTV, AB, QRV><<<grid, WAVES * WAVE, 0, stream>>>(               \
      qp, kp, vp, op, alibi_p, QH, KVH, S, LMAX, (int)kv_len, (int)kv_offset, sc, causal ? 1 : 0)
#define PF_QR(HDV, KVTV, AB)                                                                \
  do {                                                                                      \
    if (qr == 32) PF_LAUNCH(HDV, KVTV, AB, 32); else PF_LAUNCH(HDV, KVTV, AB, 16);          \
  } while (0)
  const bool ab = alibi_p != nullptr;
  if (HD == 128) {
    if (kvt_env == 32) {
      if (ab) PF_QR(128, 32, true); else PF_QR(128, 32, false);
    } else {
      if (ab) PF_QR(128, 64, true); else PF_QR(128, 64, false);
    }
  } else {
    if (kvt_env == 32) {
      if (ab) PF_QR(64, 32, true); else PF_QR(64, 32, false);
    } else {
      if (ab) PF_QR(64, 64, true); else PF_QR(64, 64, false);
    }
  }
#undef PF_QR
#undef PF_LAUNCH
  HIP_CHECK_LAST();
  return out;
}
