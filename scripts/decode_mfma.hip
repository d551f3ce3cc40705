// Prototype: MFMA decode attention (flash-decoding) for CDNA4.
//
// The shipping attn_decode_kernel is VALU-bound: each cache row costs a
// 16-lane dot-product shuffle reduce per q head, capping at ~1.5 TB/s of
// K/V traffic at kv=131k (profiles/decode_longctx_sweep.log). Here the GQ
// query heads of one kv head form the rows of an mfma_f32_16x16x32_bf16
// A-fragment (padded to 16), so 32 cache keys cost 4 QK MFMAs + 8 PV MFMAs
// per wave instead of 32*GQ shuffle-reduced dot products:
//   - K is consumed directly from global as B-fragments (no LDS staging:
//     lane (col,hi) reads K[key=col][kdim=hi*8..] - 16 B per lane);
//   - V goes through the XOR-swizzled LDS transpose (same as prefill);
//   - 4 waves take interleaved 32-key tiles of this workgroup's split range,
//     combine through LDS, then a second kernel merges splits (same
//     numerically-deterministic two-level scheme as the shipping kernel).
//
// Build: hipcc -O3 -ffast-math --offload-arch=gfx950 -o /tmp/dm scripts/decode_mfma.hip

#include "../petals_amd/ops/csrc/common.h"
#include <cmath>
#include <cstdio>
#include <vector>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define KVT 32   // keys per wave-tile
#define WAVES 4  // waves per workgroup (each on its own tile stream)
#define KPAD 8

#define VT_BYTE(dim, key_byte) \
  ((((unsigned)(dim)) * ((KVT + KPAD) * 2) + (unsigned)(key_byte)) ^ ((((unsigned)(dim) >> 3) & 7u) << 4))

template <int HD, int GQ>
__global__ __launch_bounds__(WAVES * 64) void mfma_decode_kernel(
    const float* __restrict__ q,            // [B, KV, GQ, HD]
    const unsigned short* __restrict__ k_cache,  // [B, KV, lmax, HD]
    const unsigned short* __restrict__ v_cache,
    float* __restrict__ part_o,             // [B*KV, splits, GQ, HD]
    float* __restrict__ part_ml,            // [B*KV, splits, GQ, 2]
    const int* __restrict__ kv_len_ptr,
    int kv_heads,
    int lmax,
    int n_splits,
    float scale) {
  const int bkv = blockIdx.x;
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

  // per-wave LDS: swizzled V^T tile + P scratch; cross-wave combine reuses
  // the same storage after the main loop (sized for the larger of the two)
  __shared__ unsigned char vt_raw[WAVES][HD * (KVT + KPAD) * 2];
  __shared__ unsigned short p_lds[WAVES][16][KVT + KPAD];
  __shared__ float c_ml[WAVES][GQ][2];

  // ---- q tile: A-fragment rows = q heads (zero-padded to 16)
  bf16x8 q_frag[KCH];
  const size_t q_base = (size_t)bkv * GQ * HD;
#pragma unroll
  for (int kc = 0; kc < KCH; ++kc) {
    if (col < GQ) {
      const float* src = q + q_base + (size_t)col * HD + kc * 32 + hi * 8;
      short v[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = (short)f32_to_bf16(src[e] * scale);
      q_frag[kc] = bf16x8{v[0], v[1], v[2], v[3], v[4], v[5], v[6], v[7]};
    } else {
      q_frag[kc] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
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

  const size_t kv_base = (size_t)bkv * lmax * HD;
  const unsigned short* kb = k_cache + kv_base;
  const unsigned short* vb = v_cache + kv_base;

  // wave w handles tiles j_begin + (t*WAVES + w)*KVT
  for (int j0 = j_begin + wave * KVT; j0 < j_end; j0 += WAVES * KVT) {
    const int tile_n = min(KVT, j_end - j0);
    const bool full = tile_n == KVT;

    // ---- stage V^T (this wave only; 64 lanes x 8 iters)
    for (int idx = lane; idx < KVT * (HD / 8); idx += WAVE) {
      const int row = idx / (HD / 8);
      const int c8 = (idx - row * (HD / 8)) * 8;
      bf16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (j0 + row < j_end) vv8 = *reinterpret_cast<const bf16x8*>(vb + (size_t)(j0 + row) * HD + c8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        *reinterpret_cast<unsigned short*>(&vt_raw[wave][VT_BYTE(c8 + e, row * 2)]) = (unsigned short)vv8[e];
    }
    // per-wave staging: lanes of this wave only -> no block barrier needed,
    // but LDS writes must be visible to the same wave's reads
    __builtin_amdgcn_s_waitcnt(0);

    // ---- S = Q K^T, K direct from global: B[k=kdim][n=key]
    f32x4 s_acc[KVT / 16];
#pragma unroll
    for (int nb = 0; nb < KVT / 16; ++nb) {
      s_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};
      const int key = j0 + nb * 16 + col;
      const unsigned short* krow = kb + (size_t)min(key, j_end - 1) * HD + hi * 8;
#pragma unroll
      for (int kc = 0; kc < KCH; ++kc) {
        const bf16x8 kt = *reinterpret_cast<const bf16x8*>(krow + kc * 32);
        s_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[kc], kt, s_acc[nb], 0, 0, 0);
      }
    }

    // ---- online softmax over this tile (rows = heads hi*4+r)
    float p[KVT / 16][4];
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float s[KVT / 16];
#pragma unroll
      for (int nb = 0; nb < KVT / 16; ++nb) {
        s[nb] = s_acc[nb][r];
        if (!full && j0 + nb * 16 + col >= j_end) s[nb] = NEG_SENTINEL;
      }
      float mx = NEG_SENTINEL;
#pragma unroll
      for (int nb = 0; nb < KVT / 16; ++nb) mx = fmaxf(mx, s[nb]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      const float m_new = fmaxf(m_row[r], mx);
      corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
      float lsum = 0.f;
#pragma unroll
      for (int nb = 0; nb < KVT / 16; ++nb) {
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
      for (int nb = 0; nb < KVT / 16; ++nb)
        p_lds[wave][hi * 4 + r][nb * 16 + col] = f32_to_bf16(p[nb][r]);
    __builtin_amdgcn_s_waitcnt(0);

    const bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(&p_lds[wave][col][hi * 8]);

#pragma unroll
    for (int d = 0; d < HD / 16; ++d)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[d][r] *= corr[r];
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) {
      const bf16x8 vfrag = *reinterpret_cast<const bf16x8*>(&vt_raw[wave][VT_BYTE(d * 16 + col, hi * 16)]);
      acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, vfrag, acc_o[d], 0, 0, 0);
    }
  }

  // ---- cross-wave combine through LDS (reuse p_lds area for O rows)
  // write this wave's m/l and O (C layout: row hi*4+r = head, col dims)
  __syncthreads();  // all waves done with their private LDS
  float* c_o = reinterpret_cast<float*>(vt_raw[wave]);  // [GQ][HD] f32 fits: GQ*HD*4 <= HD*(KVT+KPAD)*2
  static_assert(GQ * HD * 4 <= HD * (KVT + KPAD) * 2, "combine O overlay too big");
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int g = hi * 4 + r;
    if (g < GQ) {
#pragma unroll
      for (int d = 0; d < HD / 16; ++d) c_o[g * HD + d * 16 + col] = acc_o[d][r];
      if (col == 0) {
        c_ml[wave][g][0] = m_row[r];
        c_ml[wave][g][1] = l_row[r];
      }
    }
  }
  __syncthreads();

  // wave 0 merges the 4 waves and writes the split partial
  if (wave == 0) {
    float* po = part_o + (((size_t)bkv * n_splits + split) * GQ) * HD;
    float* pml = part_ml + (((size_t)bkv * n_splits + split) * GQ) * 2;
    for (int idx = lane; idx < GQ * HD; idx += WAVE) {
      const int g = idx / HD, d = idx - g * HD;
      float m_star = c_ml[0][g][0];
#pragma unroll
      for (int w = 1; w < WAVES; ++w) m_star = fmaxf(m_star, c_ml[w][g][0]);
      float osum = 0.f, lsum = 0.f;
#pragma unroll
      for (int w = 0; w < WAVES; ++w) {
        const float wgt = (c_ml[w][g][0] <= NEG_THRESHOLD) ? 0.f : __expf(c_ml[w][g][0] - m_star);
        osum += wgt * reinterpret_cast<const float*>(vt_raw[w])[g * HD + d];
        lsum += wgt * c_ml[w][g][1];
      }
      po[idx] = osum;
      if (d == 0) {
        pml[g * 2 + 0] = m_star;
        pml[g * 2 + 1] = lsum;
      }
    }
  }
}

// final split merge (same scheme as the shipping combine kernel)
template <int HD, int GQ>
__global__ void split_combine_kernel(
    const float* __restrict__ part_o, const float* __restrict__ part_ml,
    float* __restrict__ out, int n_splits) {
  const int bkv = blockIdx.x;
  const int g = blockIdx.y;
  const int d = threadIdx.x;
  float m_star = NEG_SENTINEL;
  for (int s = 0; s < n_splits; ++s)
    m_star = fmaxf(m_star, part_ml[(((size_t)bkv * n_splits + s) * GQ + g) * 2]);
  float osum = 0.f, lsum = 0.f;
  for (int s = 0; s < n_splits; ++s) {
    const float m = part_ml[(((size_t)bkv * n_splits + s) * GQ + g) * 2];
    const float l = part_ml[(((size_t)bkv * n_splits + s) * GQ + g) * 2 + 1];
    const float w = (m <= NEG_THRESHOLD) ? 0.f : __expf(m - m_star);
    osum += w * part_o[((((size_t)bkv * n_splits + s) * GQ) + g) * HD + d];
    lsum += w * l;
  }
  out[((size_t)bkv * GQ + g) * HD + d] = lsum > 0.f ? osum / lsum : 0.f;
}

// ---------------- host ----------------

static float bf2f(unsigned short u) {
  unsigned int x = ((unsigned int)u) << 16;
  float f;
  __builtin_memcpy(&f, &x, 4);
  return f;
}
static unsigned short f2bf(float f) {
  unsigned int x;
  __builtin_memcpy(&x, &f, 4);
  x += 0x7fff + ((x >> 16) & 1);
  return (unsigned short)(x >> 16);
}

template <int HD, int GQ>
void run_case(int B, int KV, int kv_len, int lmax, int n_splits, bool check, int iters) {
  const int BKV = B * KV;
  unsigned long long seed = 99;
  auto rnd = [&]() {
    seed = seed * 6364136223846793005ULL + 1442695040888963407ULL;
    return (float)((seed >> 33) & 0xFFFFFF) / (float)0xFFFFFF * 2.f - 1.f;
  };

  std::vector<float> q((size_t)BKV * GQ * HD);
  std::vector<unsigned short> k((size_t)BKV * lmax * HD, 0), v((size_t)BKV * lmax * HD, 0);
  for (auto& x : q) x = rnd();
  for (int bk = 0; bk < BKV; bk++)
    for (int j = 0; j < kv_len; j++)
      for (int d = 0; d < HD; d++) {
        k[((size_t)bk * lmax + j) * HD + d] = f2bf(rnd() * 0.5f);
        v[((size_t)bk * lmax + j) * HD + d] = f2bf(rnd() * 0.5f);
      }

  float *dq, *dpo, *dpml, *dout;
  unsigned short *dk, *dv;
  int* dlen;
  (void)hipMalloc(&dq, q.size() * 4);
  (void)hipMalloc(&dk, k.size() * 2);
  (void)hipMalloc(&dv, v.size() * 2);
  (void)hipMalloc(&dpo, (size_t)BKV * n_splits * GQ * HD * 4);
  (void)hipMalloc(&dpml, (size_t)BKV * n_splits * GQ * 2 * 4);
  (void)hipMalloc(&dout, (size_t)BKV * GQ * HD * 4);
  (void)hipMalloc(&dlen, 4);
  (void)hipMemcpy(dq, q.data(), q.size() * 4, hipMemcpyHostToDevice);
  (void)hipMemcpy(dk, k.data(), k.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(dv, v.data(), v.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(dlen, &kv_len, 4, hipMemcpyHostToDevice);

  const float scale = 1.0f / sqrtf((float)HD);
  dim3 grid(BKV, n_splits);
  dim3 cgrid(BKV, GQ);

  auto launch = [&]() {
    mfma_decode_kernel<HD, GQ><<<grid, WAVES * WAVE>>>(
        dq, dk, dv, dpo, dpml, dlen, KV, lmax, n_splits, scale);
    split_combine_kernel<HD, GQ><<<cgrid, HD>>>(dpo, dpml, dout, n_splits);
  };
  launch();
  (void)hipDeviceSynchronize();

  if (check) {
    std::vector<float> out((size_t)BKV * GQ * HD);
    (void)hipMemcpy(out.data(), dout, out.size() * 4, hipMemcpyDeviceToHost);
    float maxerr = 0;
    for (int bk = 0; bk < BKV; bk++)
      for (int g = 0; g < GQ; g++) {
        std::vector<float> sc(kv_len);
        float mx = -1e30f;
        for (int j = 0; j < kv_len; j++) {
          float acc = 0;
          for (int d = 0; d < HD; d++)
            acc += q[((size_t)bk * GQ + g) * HD + d] * bf2f(k[((size_t)bk * lmax + j) * HD + d]);
          sc[j] = acc * scale;
          mx = fmaxf(mx, sc[j]);
        }
        float l = 0;
        for (int j = 0; j < kv_len; j++) {
          sc[j] = expf(sc[j] - mx);
          l += sc[j];
        }
        for (int d = 0; d < HD; d++) {
          float acc = 0;
          for (int j = 0; j < kv_len; j++) acc += sc[j] * bf2f(v[((size_t)bk * lmax + j) * HD + d]);
          const float ref = acc / l;
          maxerr = fmaxf(maxerr, fabsf(out[((size_t)bk * GQ + g) * HD + d] - ref));
        }
      }
    printf("check B=%d KV=%d GQ=%d kv=%d splits=%d: max_err=%.4f %s\n", B, KV, GQ, kv_len,
           n_splits, maxerr, maxerr < 0.02f ? "OK" : "FAIL");
  } else {
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    (void)hipEventRecord(e0);
    for (int i = 0; i < iters; i++) launch();
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    ms /= iters;
    const double gb = 2.0 * BKV * kv_len * HD * 2 / 1e9;
    printf("time  B=%d KV=%d GQ=%d kv=%6d splits=%3d: %8.1f us  %5.0f GB/s\n", B, KV, GQ, kv_len,
           n_splits, ms * 1e3, gb / (ms * 1e-3));
  }
  (void)hipFree(dq); (void)hipFree(dk); (void)hipFree(dv);
  (void)hipFree(dpo); (void)hipFree(dpml); (void)hipFree(dout); (void)hipFree(dlen);
}

int main() {
  // numerics: odd kv_len (tail masking), multiple splits, GQ 8 and 4 and 16
  run_case<128, 8>(2, 2, 203, 256, 3, true, 0);
  run_case<128, 8>(1, 3, 1000, 1024, 7, true, 0);
  run_case<128, 4>(1, 2, 77, 128, 2, true, 0);
  run_case<128, 16>(1, 2, 500, 512, 4, true, 0);
  run_case<64, 8>(1, 2, 300, 512, 3, true, 0);
  // timing at the llama-2-70b shape
  for (int kv : {4096, 16384, 32768, 131072}) {
    for (int splits : {32, 64, 96, 128}) {
      run_case<128, 8>(1, 8, kv, kv, splits, false, 20);
    }
  }
  return 0;
}
