// Standalone bisect harness for the prefill-attention numerics failure
// (dead keys leak into C-layout rows with reg index r=0).
//
// Variants, all KVT-templated:
//   0 = exact copy of the current ops/csrc/prefill_attn.hip kernel
//   1 = exact copy of the OLD (pre-KVT) kernel, hardwired NB=2 scalar softmax
//   2 = new kernel, but mask+mx split into two loops
//   3 = new kernel, but p[][] written in the same loop that computes expf
//       replaced by old-style explicit scalar temporaries (NB=2 only)
//
// Build: hipcc -O3 -ffast-math --offload-arch=gfx950 -o /tmp/pb scripts/prefill_bisect.hip
// Run:   /tmp/pb  (prints max err per variant vs CPU fp32 reference)

#include "../petals_amd/ops/csrc/common.h"
#include <cmath>
#include <cstdio>
#include <vector>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

#define QTILE 16
#define WAVES 4
#define KPAD 8

#define VT_BYTE(dim, key_byte) \
  ((((unsigned)(dim)) * ((KVT + KPAD) * 2) + (unsigned)(key_byte)) ^ ((((unsigned)(dim) >> 3) & 7u) << 4))

template <int HD, int KVT, int VAR>
__global__ __launch_bounds__(WAVES * 64) void kern(
    const unsigned short* __restrict__ q, const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v, unsigned short* __restrict__ out,
    int q_heads, int kv_heads, int s_q, int lmax, int kv_len, int kv_offset,
    float scale, int causal) {
  const int bh = blockIdx.x;
  const int b = bh / q_heads;
  const int qh = bh - b * q_heads;
  const int kvh = qh / (q_heads / kv_heads);
  const int q0_wg = blockIdx.y * (WAVES * QTILE);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int hi = lane >> 4;
  const int q0 = q0_wg + wave * QTILE;

  __shared__ unsigned short k_lds[KVT][HD + KPAD];
  __shared__ unsigned char vt_raw[HD * (KVT + KPAD) * 2];
  __shared__ unsigned char p_raw[WAVES * QTILE * (KVT + KPAD) * 2];
  constexpr bool PSWZ = (VAR == 4);
#define P_BYTE(wv, row, kb)                                                     \
  ((((unsigned)(wv)*QTILE + (unsigned)(row)) * ((KVT + KPAD) * 2) + (unsigned)(kb)) ^ \
   (PSWZ ? ((((unsigned)(row) >> 2) & 3u) << 4) : 0u))
#define p_lds_write(wv, row, colidx, val) \
  (*reinterpret_cast<unsigned short*>(&p_raw[P_BYTE(wv, row, (colidx)*2)]) = (val))

  constexpr int KCH = HD / 32;
  constexpr int NB = KVT / 16;
  constexpr int PKC = KVT / 32;

  bf16x8 q_frag[KCH];
  const size_t q_base = (((size_t)b * q_heads + qh) * s_q) * HD;
  const int my_qrow = q0 + col;
#pragma unroll
  for (int kc = 0; kc < KCH; ++kc) {
    if (my_qrow < s_q) {
      const unsigned short* src = q + q_base + (size_t)my_qrow * HD + kc * 32 + hi * 8;
      q_frag[kc] = *reinterpret_cast<const bf16x8*>(src);
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

  const int wg_last_q_abs = kv_offset + min(q0_wg + WAVES * QTILE, s_q) - 1;
  const int kv_end = causal ? min(kv_len, wg_last_q_abs + 1) : kv_len;
  const size_t kv_base = (((size_t)b * kv_heads + kvh) * lmax) * HD;

  constexpr int SIT = (KVT * (HD / 8) + WAVES * WAVE - 1) / (WAVES * WAVE);  // staging iters/thread
  bf16x8 k_reg[SIT], v_reg[SIT];
  if constexpr (VAR == 6) {
    // preload tile 0 into registers
#pragma unroll
    for (int it = 0; it < SIT; ++it) {
      const int idx = tid + it * WAVES * WAVE;
      const int row = idx / (HD / 8);
      const int c8 = (idx - row * (HD / 8)) * 8;
      k_reg[it] = v_reg[it] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      if (idx < KVT * (HD / 8) && row < kv_end) {
        k_reg[it] = *reinterpret_cast<const bf16x8*>(k + kv_base + (size_t)row * HD + c8);
        v_reg[it] = *reinterpret_cast<const bf16x8*>(v + kv_base + (size_t)row * HD + c8);
      }
    }
  }

  for (int j0 = 0; j0 < kv_end; j0 += KVT) {
    const int tile_n = min(KVT, kv_end - j0);
    __syncthreads();
    if constexpr (VAR == 6) {
      // write the prefetched tile, then start the NEXT tile's global loads so
      // they overlap the MFMA/softmax compute below
#pragma unroll
      for (int it = 0; it < SIT; ++it) {
        const int idx = tid + it * WAVES * WAVE;
        if (idx < KVT * (HD / 8)) {
          const int row = idx / (HD / 8);
          const int c8 = (idx - row * (HD / 8)) * 8;
          *reinterpret_cast<bf16x8*>(&k_lds[row][c8]) = k_reg[it];
#pragma unroll
          for (int e = 0; e < 8; ++e)
            *reinterpret_cast<unsigned short*>(&vt_raw[VT_BYTE(c8 + e, row * 2)]) = (unsigned short)v_reg[it][e];
        }
      }
      __syncthreads();
      if (j0 + KVT < kv_end) {
        const int jn = j0 + KVT;
#pragma unroll
        for (int it = 0; it < SIT; ++it) {
          const int idx = tid + it * WAVES * WAVE;
          const int row = idx / (HD / 8);
          const int c8 = (idx - row * (HD / 8)) * 8;
          k_reg[it] = v_reg[it] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
          if (idx < KVT * (HD / 8) && jn + row < kv_end) {
            k_reg[it] = *reinterpret_cast<const bf16x8*>(k + kv_base + (size_t)(jn + row) * HD + c8);
            v_reg[it] = *reinterpret_cast<const bf16x8*>(v + kv_base + (size_t)(jn + row) * HD + c8);
          }
        }
      }
    } else {
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
    }

    f32x4 s_acc[NB];
#pragma unroll
    for (int nb = 0; nb < NB; ++nb) {
      s_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < KCH; ++kc) {
        const bf16x8 kt = *reinterpret_cast<const bf16x8*>(&k_lds[nb * 16 + col][kc * 32 + hi * 8]);
        s_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[kc], kt, s_acc[nb], 0, 0, 0);
      }
    }

    float p[NB][4];
    float corr[4];
    const bool interior =
        (tile_n == KVT) && (q0 + QTILE <= s_q) && (!causal || j0 + KVT <= kv_offset + q0 + 1);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + hi * 4 + r;
      const int q_abs = kv_offset + qrow;

      if constexpr (VAR == 1 || VAR == 3) {
        // old scalar softmax (NB==2 only)
        static_assert(NB == 2, "scalar variants are KVT=32 only");
        float s0 = s_acc[0][r] * scale;
        float s1 = s_acc[1][r] * scale;
        const int key0 = j0 + col, key1 = j0 + 16 + col;
        const bool dead0 = key0 >= tile_n + j0 || (causal && key0 > q_abs) || qrow >= s_q;
        const bool dead1 = key1 >= tile_n + j0 || (causal && key1 > q_abs) || qrow >= s_q;
        if (dead0) s0 = NEG_SENTINEL;
        if (dead1) s1 = NEG_SENTINEL;
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
      } else if constexpr (VAR == 5 || VAR == 6) {
        // split-mask form + mask-free interior tiles (all rows of this wave
        // live for the whole tile: no tail, no causal edge, no s_q edge)
        float s[NB];
        if (interior) {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) s[nb] = s_acc[nb][r] * scale;
        } else {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) {
            s[nb] = s_acc[nb][r] * scale;
            const int key = j0 + nb * 16 + col;
            if (key >= tile_n + j0 || (causal && key > q_abs) || qrow >= s_q) s[nb] = NEG_SENTINEL;
          }
        }
        float mx = NEG_SENTINEL;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) mx = fmaxf(mx, s[nb]);
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_row[r], mx);
        corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
        float lsum = 0.f;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          p[nb][r] = (s[nb] <= NEG_THRESHOLD) ? 0.f : __expf(s[nb] - m_new);
          lsum += p[nb][r];
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_row[r] = l_row[r] * corr[r] + lsum;
        m_row[r] = m_new;
      } else if constexpr (VAR == 2 || VAR == 4) {
        // split: mask loop, then mx loop
        float s[NB];
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          s[nb] = s_acc[nb][r] * scale;
          const int key = j0 + nb * 16 + col;
          if (key >= tile_n + j0 || (causal && key > q_abs) || qrow >= s_q) s[nb] = NEG_SENTINEL;
        }
        float mx = NEG_SENTINEL;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) mx = fmaxf(mx, s[nb]);
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_row[r], mx);
        corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
        float lsum = 0.f;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          p[nb][r] = (s[nb] <= NEG_THRESHOLD) ? 0.f : __expf(s[nb] - m_new);
          lsum += p[nb][r];
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_row[r] = l_row[r] * corr[r] + lsum;
        m_row[r] = m_new;
      } else {
        // VAR == 0: exact current kernel
        float s[NB];
        float mx = NEG_SENTINEL;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          s[nb] = s_acc[nb][r] * scale;
          const int key = j0 + nb * 16 + col;
          const bool dead = key >= tile_n + j0 || (causal && key > q_abs) || qrow >= s_q;
          if (dead) s[nb] = NEG_SENTINEL;
          mx = fmaxf(mx, s[nb]);
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        const float m_new = fmaxf(m_row[r], mx);
        corr[r] = (m_row[r] <= NEG_THRESHOLD) ? 0.f : __expf(m_row[r] - m_new);
        float lsum = 0.f;
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          p[nb][r] = (s[nb] <= NEG_THRESHOLD) ? 0.f : __expf(s[nb] - m_new);
          lsum += p[nb][r];
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_row[r] = l_row[r] * corr[r] + lsum;
        m_row[r] = m_new;
      }
    }

    if constexpr (VAR == 1) {
      // old write form
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds_write(wave, hi * 4 + r, col, f32_to_bf16(p[0][r]));
        p_lds_write(wave, hi * 4 + r, 16 + col, f32_to_bf16(p[1][r]));
      }
    } else {
#pragma unroll
      for (int r = 0; r < 4; ++r)
#pragma unroll
        for (int nb = 0; nb < NB; ++nb)
          p_lds_write(wave, hi * 4 + r, nb * 16 + col, f32_to_bf16(p[nb][r]));
    }
    __syncthreads();

    bf16x8 p_frag[PKC];
#pragma unroll
    for (int pk = 0; pk < PKC; ++pk)
      p_frag[pk] = *reinterpret_cast<const bf16x8*>(&p_raw[P_BYTE(wave, col, pk * 64 + hi * 16)]);

#pragma unroll
    for (int d = 0; d < HD / 16; ++d)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[d][r] *= corr[r];
#pragma unroll
    for (int d = 0; d < HD / 16; ++d) {
#pragma unroll
      for (int pk = 0; pk < PKC; ++pk) {
        const bf16x8 vfrag =
            *reinterpret_cast<const bf16x8*>(&vt_raw[VT_BYTE(d * 16 + col, pk * 64 + hi * 16)]);
        acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag[pk], vfrag, acc_o[d], 0, 0, 0);
      }
    }
  }

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
  unsigned int lsb = (x >> 16) & 1;
  x += 0x7fff + lsb;
  return (unsigned short)(x >> 16);
}

int main() {
  const int B = 1, QH = 2, KVH = 2, S = 64, HD = 128;
  const int kv_len = S, lmax = S + 16, off = 0;
  const int causal = 1;
  const float scale = 1.0f / sqrtf((float)HD);

  unsigned long long seed = 12345;
  auto rnd = [&]() {
    seed = seed * 6364136223846793005ULL + 1442695040888963407ULL;
    return (float)((seed >> 33) & 0xFFFFFF) / (float)0xFFFFFF * 2.f - 1.f;
  };

  std::vector<unsigned short> q(B * QH * S * HD), k(B * KVH * lmax * HD, 0), v(B * KVH * lmax * HD, 0);
  for (auto& x : q) x = f2bf(rnd() * 0.5f);
  for (int b = 0; b < B; b++)
    for (int h = 0; h < KVH; h++)
      for (int j = 0; j < kv_len; j++)
        for (int d = 0; d < HD; d++) {
          k[((b * KVH + h) * (size_t)lmax + j) * HD + d] = f2bf(rnd() * 0.5f);
          v[((b * KVH + h) * (size_t)lmax + j) * HD + d] = f2bf(rnd() * 0.5f);
        }

  // CPU fp32 reference
  std::vector<float> ref(B * QH * S * HD);
  for (int b = 0; b < B; b++)
    for (int h = 0; h < QH; h++) {
      const int kvh = h / (QH / KVH);
      for (int i = 0; i < S; i++) {
        std::vector<float> sc(kv_len);
        float mx = -1e30f;
        for (int j = 0; j < kv_len; j++) {
          float acc = 0;
          for (int d = 0; d < HD; d++)
            acc += bf2f(q[((b * QH + h) * (size_t)S + i) * HD + d]) *
                   bf2f(k[((b * KVH + kvh) * (size_t)lmax + j) * HD + d]);
          acc *= scale;
          if (causal && j > off + i) acc = -1e30f;
          sc[j] = acc;
          mx = fmaxf(mx, acc);
        }
        float l = 0;
        for (int j = 0; j < kv_len; j++) {
          sc[j] = (sc[j] <= -1e29f) ? 0.f : expf(sc[j] - mx);
          l += sc[j];
        }
        for (int d = 0; d < HD; d++) {
          float acc = 0;
          for (int j = 0; j < kv_len; j++)
            acc += sc[j] * bf2f(v[((b * KVH + kvh) * (size_t)lmax + j) * HD + d]);
          ref[((b * QH + h) * (size_t)S + i) * HD + d] = acc / l;
        }
      }
    }

  unsigned short *dq, *dk, *dv, *dout;
  (void)hipMalloc(&dq, q.size() * 2);
  (void)hipMalloc(&dk, k.size() * 2);
  (void)hipMalloc(&dv, v.size() * 2);
  (void)hipMalloc(&dout, q.size() * 2);
  (void)hipMemcpy(dq, q.data(), q.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(dk, k.data(), k.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(dv, v.data(), v.size() * 2, hipMemcpyHostToDevice);

  dim3 grid(B * QH, (S + 63) / 64);
  std::vector<unsigned short> out(q.size());

  auto check = [&](const char* name) {
    (void)hipMemcpy(out.data(), dout, out.size() * 2, hipMemcpyDeviceToHost);
    float err = 0;
    int bad_rows[16], nbad = 0;
    for (int i = 0; i < S && nbad < 16; i++) {
      float rowerr = 0;
      for (int h = 0; h < QH; h++)
        for (int d = 0; d < HD; d++) {
          float e = fabsf(bf2f(out[(h * (size_t)S + i) * HD + d]) - ref[(h * (size_t)S + i) * HD + d]);
          rowerr = fmaxf(rowerr, e);
        }
      err = fmaxf(err, rowerr);
      if (rowerr > 0.05f) bad_rows[nbad++] = i;
    }
    printf("%-28s max_err=%.4f bad_rows:", name, err);
    for (int i = 0; i < nbad; i++) printf(" %d", bad_rows[i]);
    printf("\n");
  };

#define RUN(HD_, KVT_, VAR_, NAME)                                                         \
  (void)hipMemset(dout, 0, out.size() * 2);                                                \
  kern<HD_, KVT_, VAR_><<<grid, WAVES * WAVE>>>(dq, dk, dv, dout, QH, KVH, S, lmax,        \
                                                kv_len, off, scale, causal);               \
  (void)hipDeviceSynchronize();                                                            \
  check(NAME);

  RUN(128, 32, 0, "KVT32 VAR0 (new exact)")
  RUN(128, 32, 1, "KVT32 VAR1 (old exact)")
  RUN(128, 32, 2, "KVT32 VAR2 (split mask/mx)")
  RUN(128, 32, 3, "KVT32 VAR3 (old smax/new wr)")
  RUN(128, 32, 4, "KVT32 VAR4 (split + P swz)")
  RUN(128, 64, 0, "KVT64 VAR0 (new exact)")
  RUN(128, 64, 2, "KVT64 VAR2 (split mask/mx)")
  RUN(128, 64, 4, "KVT64 VAR4 (split + P swz)")
  RUN(128, 64, 5, "KVT64 VAR5 (interior fast)")
  RUN(128, 64, 6, "KVT64 VAR6 (prefetch)")
  RUN(128, 32, 6, "KVT32 VAR6 (prefetch)")

  // ---- timing at the llama-2-70b prefill shape ----
  {
    const int B2 = 1, QH2 = 64, KVH2 = 8, S2 = 4096, HD2 = 128;
    unsigned short *tq, *tk, *tv, *tout;
    (void)hipMalloc(&tq, (size_t)B2 * QH2 * S2 * HD2 * 2);
    (void)hipMalloc(&tk, (size_t)B2 * KVH2 * S2 * HD2 * 2);
    (void)hipMalloc(&tv, (size_t)B2 * KVH2 * S2 * HD2 * 2);
    (void)hipMalloc(&tout, (size_t)B2 * QH2 * S2 * HD2 * 2);
    (void)hipMemset(tq, 0x3c, (size_t)B2 * QH2 * S2 * HD2 * 2);
    (void)hipMemset(tk, 0x3c, (size_t)B2 * KVH2 * S2 * HD2 * 2);
    (void)hipMemset(tv, 0x3c, (size_t)B2 * KVH2 * S2 * HD2 * 2);
    dim3 tg(B2 * QH2, (S2 + 63) / 64);
    const double flops = 4.0 * B2 * QH2 * (double)S2 * S2 * HD2 / 2.0;

#define TIME(KVT_, VAR_, NAME)                                                            \
    {                                                                                     \
      for (int i = 0; i < 3; i++)                                                         \
        kern<128, KVT_, VAR_><<<tg, WAVES * WAVE>>>(tq, tk, tv, tout, QH2, KVH2, S2, S2,  \
                                                    S2, 0, 0.0883883f, 1);                \
      (void)hipDeviceSynchronize();                                                       \
      hipEvent_t e0, e1;                                                                  \
      (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);                               \
      (void)hipEventRecord(e0);                                                           \
      for (int i = 0; i < 20; i++)                                                        \
        kern<128, KVT_, VAR_><<<tg, WAVES * WAVE>>>(tq, tk, tv, tout, QH2, KVH2, S2, S2,  \
                                                    S2, 0, 0.0883883f, 1);                \
      (void)hipEventRecord(e1);                                                           \
      (void)hipEventSynchronize(e1);                                                      \
      float ms = 0; (void)hipEventElapsedTime(&ms, e0, e1); ms /= 20.f;                   \
      printf("%-28s %.3f ms  %.0f TF\n", NAME, ms, flops / (ms * 1e-3) / 1e12);          \
    }

    TIME(32, 2, "time KVT32 split")
    TIME(32, 4, "time KVT32 split+Pswz")
    TIME(64, 2, "time KVT64 split")
    TIME(64, 4, "time KVT64 split+Pswz")
    TIME(64, 5, "time KVT64 interior-fast")
    TIME(64, 6, "time KVT64 prefetch")
  }
  return 0;
}
