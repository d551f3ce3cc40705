// Device-routed MoE expert GEMVs for CDNA4 (gfx950).
//
// Decode-time grouped GEMM: virtual row r = (token b = r/K, slot j = r%K)
// reads its expert id from a DEVICE tensor (written by the on-GPU router:
// softmax + top-k, no host sync) and streams that expert's stacked weights.
// Everything is hipGraph-capturable — the expert choice is data in device
// buffers, not launch parameters — which is what lets Mixtral spans use the
// whole-span decode graph (the reference runs Mixtral experts densely inside
// HF code with host-side routing: reference models/mixtral/block.py:73-81).
//
// Weight layouts match gemv.hip / nf4.hip exactly, stacked on a leading
// expert axis:
//   bf16: wt_all   [E, in, out]
//   nf4:  packed   [E, in, out/2], absmax [E, in, out/64]
// The per-row inner loop is the same x-broadcast split-K structure as the
// dense kernels (each selected expert's weights are read exactly once per
// token).

#include "common.h"
#include "gemv_reduce.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#ifndef GEMV_OUT_PER_WAVE
#define GEMV_OUT_PER_WAVE 512
#endif
#define NF4_OUT_PER_WAVE 1024

using bf16x8_moe = __attribute__((ext_vector_type(8))) short;
using f32x4_moe = __attribute__((ext_vector_type(4))) float;

static __device__ __constant__ float MOE_NF4_LUT[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f, -0.39491748809814453f,
    -0.28444138169288635f, -0.18477343022823334f, -0.09105003625154495f, 0.0f,
    0.07958029955625534f, 0.16093020141124725f, 0.24611230194568634f,
    0.33791524171829224f, 0.4407098293304443f, 0.5626170039176941f,
    0.7229568362236023f, 1.0f};

// ------------------------------------------------------------ bf16 indexed

__global__ __launch_bounds__(WAVE) void gemv_bf16_moe_kernel(
    const unsigned short* __restrict__ wt_all,  // [E, in, out]
    const float* __restrict__ x,                // [B, in]
    const int* __restrict__ sel,                // [R] expert per virtual row
    float* __restrict__ partials,               // [n_splits, R, out]
    int in_dim,
    int out_dim,
    int i_per_split,
    int rows,
    int k_per_tok) {
  const int row = blockIdx.z;
  const int lane = threadIdx.x & (WAVE - 1);
  const int out0 = blockIdx.x * GEMV_OUT_PER_WAVE + lane * 8;
  if (out0 >= out_dim) return;
  const int split = blockIdx.y;
  const int i_begin = split * i_per_split;
  const int i_end = min(i_begin + i_per_split, in_dim);

  const int e = sel[row];
  const unsigned short* wt = wt_all + (size_t)e * in_dim * out_dim;
  const float* xr = x + (size_t)(row / k_per_tok) * in_dim;

  float acc[8];
#pragma unroll
  for (int v = 0; v < 8; ++v) acc[v] = 0.f;

  const bool full = (out0 + 8) <= out_dim;
  if (full) {
    constexpr int UNROLL = 16;
    const unsigned short* wp = wt + (size_t)i_begin * out_dim + out0;
    int i = i_begin;
    for (; i + UNROLL <= i_end; i += UNROLL) {
      short8 w8[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        w8[u] = *reinterpret_cast<const short8*>(wp + (size_t)u * out_dim);
      float xs[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) xs[u] = xr[i + u];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
#pragma unroll
        for (int v = 0; v < 8; ++v)
          acc[v] = fmaf(bf16_to_f32((unsigned short)w8[u][v]), xs[u], acc[v]);
      }
      wp += (size_t)UNROLL * out_dim;
    }
    for (; i < i_end; ++i) {
      const short8 w8 = *reinterpret_cast<const short8*>(wt + (size_t)i * out_dim + out0);
      const float xv = xr[i];
#pragma unroll
      for (int v = 0; v < 8; ++v) acc[v] = fmaf(bf16_to_f32((unsigned short)w8[v]), xv, acc[v]);
    }
  } else {
    for (int i = i_begin; i < i_end; ++i) {
      const float xv = xr[i];
#pragma unroll
      for (int v = 0; v < 8; ++v)
        if (out0 + v < out_dim)
          acc[v] = fmaf(bf16_to_f32(wt[(size_t)i * out_dim + out0 + v]), xv, acc[v]);
    }
  }

  float* dst = partials + ((size_t)split * rows + row) * out_dim + out0;
  if (full) {
    float4v* d4 = reinterpret_cast<float4v*>(dst);
    d4[0] = float4v{acc[0], acc[1], acc[2], acc[3]};
    d4[1] = float4v{acc[4], acc[5], acc[6], acc[7]};
  } else {
#pragma unroll
    for (int v = 0; v < 8; ++v)
      if (out0 + v < out_dim) dst[v] = acc[v];
  }
}

// ------------------------------------------------------------- nf4 indexed

__global__ __launch_bounds__(WAVE) void gemv_nf4_moe_kernel(
    const unsigned char* __restrict__ packed_all,   // [E, in, out/2]
    const unsigned short* __restrict__ absmax_all,  // [E, in, out/64]
    const float* __restrict__ x,                    // [B, in]
    const int* __restrict__ sel,                    // [R]
    float* __restrict__ partials,                   // [n_splits, R, out]
    int in_dim,
    int out_dim,
    int i_per_split,
    int rows,
    int k_per_tok) {
  __shared__ float2 lut2[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    lut2[i] = make_float2(MOE_NF4_LUT[i & 0xF], MOE_NF4_LUT[i >> 4]);
  __syncthreads();

  const int row = blockIdx.z;
  const int lane = threadIdx.x & (WAVE - 1);
  const int out0 = blockIdx.x * NF4_OUT_PER_WAVE + lane * 16;
  if (out0 >= out_dim) return;
  const int split = blockIdx.y;
  const int i_begin = split * i_per_split;
  const int i_end = min(i_begin + i_per_split, in_dim);
  const bool full = (out0 + 16) <= out_dim;

  const int e = sel[row];
  const unsigned char* packed = packed_all + (size_t)e * in_dim * (out_dim >> 1);
  const unsigned short* absmax = absmax_all + (size_t)e * in_dim * (out_dim >> 6);
  const float* xr = x + (size_t)(row / k_per_tok) * in_dim;

  float acc[16];
#pragma unroll
  for (int v = 0; v < 16; ++v) acc[v] = 0.f;

  if (full) {
    constexpr int UNROLL = 16;
    const int half_out = out_dim >> 1;
    const unsigned char* pp = packed + (size_t)i_begin * half_out + (out0 >> 1);
    int i = i_begin;
    for (; i + UNROLL <= i_end; i += UNROLL) {
      uint2 pk[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        pk[u] = *reinterpret_cast<const uint2*>(pp + (size_t)u * half_out);
      float am[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u)
        am[u] = bf16_to_f32(absmax[(size_t)(i + u) * (out_dim >> 6) + (out0 >> 6)]);
      float xs[UNROLL];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) xs[u] = xr[i + u];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        const float xa = xs[u] * am[u];
        const unsigned int wds2[2] = {pk[u].x, pk[u].y};
#pragma unroll
        for (int d = 0; d < 2; ++d)
#pragma unroll
          for (int p2 = 0; p2 < 4; ++p2) {
            const float2 w2 = lut2[(wds2[d] >> (8 * p2)) & 0xFFu];
            acc[8 * d + 2 * p2] = fmaf(w2.x, xa, acc[8 * d + 2 * p2]);
            acc[8 * d + 2 * p2 + 1] = fmaf(w2.y, xa, acc[8 * d + 2 * p2 + 1]);
          }
      }
      pp += (size_t)UNROLL * half_out;
    }
    for (; i < i_end; ++i) {
      const uint2 pk = *reinterpret_cast<const uint2*>(packed + (size_t)i * half_out + (out0 >> 1));
      const float xa = xr[i] * bf16_to_f32(absmax[(size_t)i * (out_dim >> 6) + (out0 >> 6)]);
      const unsigned int twds[2] = {pk.x, pk.y};
#pragma unroll
      for (int d = 0; d < 2; ++d)
#pragma unroll
        for (int p2 = 0; p2 < 4; ++p2) {
          const float2 w2 = lut2[(twds[d] >> (8 * p2)) & 0xFFu];
          acc[8 * d + 2 * p2] = fmaf(w2.x, xa, acc[8 * d + 2 * p2]);
          acc[8 * d + 2 * p2 + 1] = fmaf(w2.y, xa, acc[8 * d + 2 * p2 + 1]);
        }
    }
  } else {
    for (int i = i_begin; i < i_end; ++i) {
      const float xa = xr[i] * bf16_to_f32(absmax[(size_t)i * (out_dim >> 6) + (out0 >> 6)]);
#pragma unroll
      for (int v = 0; v < 16; ++v)
        if (out0 + v < out_dim) {
          const unsigned char byte = packed[(size_t)i * (out_dim >> 1) + ((out0 + v) >> 1)];
          const float2 w2 = lut2[byte];
          acc[v] = fmaf(((out0 + v) & 1) ? w2.y : w2.x, xa, acc[v]);
        }
    }
  }

  float* dst = partials + ((size_t)split * rows + row) * out_dim + out0;
  if (full) {
#pragma unroll
    for (int q = 0; q < 4; ++q)
      reinterpret_cast<float4v*>(dst)[q] =
          float4v{acc[4 * q], acc[4 * q + 1], acc[4 * q + 2], acc[4 * q + 3]};
  } else {
#pragma unroll
    for (int v = 0; v < 16; ++v)
      if (out0 + v < out_dim) dst[v] = acc[v];
  }
}

// ------------------------------------------------- grouped GEMM (prefill)
//
// Prefill-time grouped GEMM over expert-sorted (token, slot) pairs: the host
// (fused_moe.py) sorts the R = T*k pairs by expert with each expert's segment
// padded to a multiple of MT rows, so every MT-row tile belongs to ONE expert
// (tile_expert[tile], -1 = pure padding). Each workgroup computes an
// MT x NT C tile with mfma_f32_16x16x32_bf16: the expert's B tile (KT x NT)
// is staged per k-step into LDS TRANSPOSED with the attention kernel's XOR
// swizzle (prefill_attn.hip VT_BYTE) so B-fragments are contiguous
// ds_read_b128 rows; NF4 weights are dequantized on the way into LDS, so the
// packed form (4.25 b/param) is what crosses HBM — no dense dequant pass.
// Replaces the per-expert dense-matmul loop for prefill (each expert's
// weights are read ceil(rows_e/MT) times instead of once per token).

#define MOE_MT 32        // C tile rows (2 MFMA row blocks)
#define MOE_NT 128       // C tile cols (4 waves x 2 MFMA col blocks)
#define MOE_KT 64        // k (in-dim) step staged in LDS
#define MOE_KPAD 8       // LDS k padding (elements)

#define BT_BYTE(n, k_byte) \
  ((((unsigned)(n)) * ((MOE_KT + MOE_KPAD) * 2) + (unsigned)(k_byte)) ^ ((((unsigned)(n) >> 3) & 7u) << 4))

template <bool NF4>
__global__ __launch_bounds__(256) void moe_gemm_kernel(
    const unsigned short* __restrict__ wt_all,      // [E, in, out] bf16 (NF4: null)
    const unsigned char* __restrict__ packed_all,   // [E, in, out/2] u8 (bf16: null)
    const unsigned short* __restrict__ absmax_all,  // [E, in, out/64] bf16 (bf16: null)
    const unsigned short* __restrict__ x,           // [T, in] bf16
    const int* __restrict__ sorted_pairs,           // [Rp] pair id or -1 (padding)
    const int* __restrict__ tile_expert,            // [Rp/MT] expert id or -1
    unsigned short* __restrict__ c,                 // [R, out] bf16
    int in_dim,
    int out_dim,
    int k_per_tok) {
  const int tile = blockIdx.x;
  const int e = tile_expert[tile];
  if (e < 0) return;  // pure-padding tile
  const int n0 = blockIdx.y * MOE_NT;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int hi = lane >> 4;

  __shared__ float2 lut2[256];
  if (NF4) {
    for (int i = tid; i < 256; i += 256)
      lut2[i] = make_float2(MOE_NF4_LUT[i & 0xF], MOE_NF4_LUT[i >> 4]);
  }
  __shared__ unsigned char bt_raw[MOE_NT * (MOE_KT + MOE_KPAD) * 2];

  // this lane's A rows (2 row blocks): token index per MFMA A row
  int tok[2];
  int cpair[2][4];  // C rows: pair ids for rows rb*16 + hi*4 + r
#pragma unroll
  for (int rb = 0; rb < 2; ++rb) {
    const int pr = sorted_pairs[tile * MOE_MT + rb * 16 + col];
    tok[rb] = pr >= 0 ? pr / k_per_tok : -1;
#pragma unroll
    for (int r = 0; r < 4; ++r)
      cpair[rb][r] = sorted_pairs[tile * MOE_MT + rb * 16 + hi * 4 + r];
  }

  f32x4_moe acc[2][2];
#pragma unroll
  for (int rb = 0; rb < 2; ++rb)
#pragma unroll
    for (int nb = 0; nb < 2; ++nb) acc[rb][nb] = f32x4_moe{0.f, 0.f, 0.f, 0.f};

  const size_t w_stride = (size_t)in_dim * out_dim;

  for (int kt = 0; kt < in_dim; kt += MOE_KT) {
    // ---- stage B tile [MOE_KT k][MOE_NT n] transposed+swizzled into LDS
    __syncthreads();
    {
      const int krow = kt + (tid >> 2);          // 64 rows, 4 threads per row
      const int nloc = (tid & 3) * 32;           // 32 n values per thread
      if (NF4) {
        // 16 packed bytes -> 32 values; one absmax block per thread
        unsigned int pk4[4] = {0u, 0u, 0u, 0u};
        float am = 0.f;
        if (krow < in_dim && n0 + nloc < out_dim) {
          const unsigned char* prow =
              packed_all + (size_t)e * (w_stride >> 1) + (size_t)krow * (out_dim >> 1) + ((n0 + nloc) >> 1);
          const uint4 p = *reinterpret_cast<const uint4*>(prow);
          pk4[0] = p.x; pk4[1] = p.y; pk4[2] = p.z; pk4[3] = p.w;
          am = bf16_to_f32(absmax_all[(size_t)e * (w_stride >> 6) + (size_t)krow * (out_dim >> 6) + ((n0 + nloc) >> 6)]);
        }
        const int kb = (krow - kt) * 2;  // k byte offset in LDS rows
#pragma unroll
        for (int w4 = 0; w4 < 4; ++w4)
#pragma unroll
          for (int b = 0; b < 4; ++b) {
            const float2 w2 = lut2[(pk4[w4] >> (8 * b)) & 0xFFu];
            const int n = nloc + w4 * 8 + b * 2;
            *reinterpret_cast<unsigned short*>(&bt_raw[BT_BYTE(n, kb)]) = f32_to_bf16(w2.x * am);
            *reinterpret_cast<unsigned short*>(&bt_raw[BT_BYTE(n + 1, kb)]) = f32_to_bf16(w2.y * am);
          }
      } else {
        const int kb = (krow - kt) * 2;
#pragma unroll
        for (int c8 = 0; c8 < 4; ++c8) {
          short8 w8 = {0, 0, 0, 0, 0, 0, 0, 0};
          if (krow < in_dim && n0 + nloc + c8 * 8 < out_dim)
            w8 = *reinterpret_cast<const short8*>(
                wt_all + (size_t)e * w_stride + (size_t)krow * out_dim + n0 + nloc + c8 * 8);
#pragma unroll
          for (int v = 0; v < 8; ++v)
            *reinterpret_cast<unsigned short*>(&bt_raw[BT_BYTE(nloc + c8 * 8 + v, kb)]) =
                (unsigned short)w8[v];
        }
      }
    }
    __syncthreads();

    // ---- MFMA: 2 k-chunks x 2 row blocks x 2 col blocks
#pragma unroll
    for (int kc = 0; kc < MOE_KT / 32; ++kc) {
      bf16x8_moe a_frag[2];
#pragma unroll
      for (int rb = 0; rb < 2; ++rb) {
        if (tok[rb] >= 0 && kt + kc * 32 + hi * 8 + 8 <= in_dim) {
          a_frag[rb] = *reinterpret_cast<const bf16x8_moe*>(
              x + (size_t)tok[rb] * in_dim + kt + kc * 32 + hi * 8);
        } else if (tok[rb] >= 0 && kt + kc * 32 + hi * 8 < in_dim) {
          short vs[8] = {0, 0, 0, 0, 0, 0, 0, 0};
          for (int v = 0; v < 8 && kt + kc * 32 + hi * 8 + v < in_dim; ++v)
            vs[v] = (short)x[(size_t)tok[rb] * in_dim + kt + kc * 32 + hi * 8 + v];
          a_frag[rb] = bf16x8_moe{vs[0], vs[1], vs[2], vs[3], vs[4], vs[5], vs[6], vs[7]};
        } else {
          a_frag[rb] = bf16x8_moe{0, 0, 0, 0, 0, 0, 0, 0};
        }
      }
#pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        const int n = wave * 32 + nb * 16 + col;
        const bf16x8_moe b_frag =
            *reinterpret_cast<const bf16x8_moe*>(&bt_raw[BT_BYTE(n, kc * 64 + hi * 16)]);
#pragma unroll
        for (int rb = 0; rb < 2; ++rb)
          acc[rb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[rb], b_frag, acc[rb][nb], 0, 0, 0);
      }
    }
  }

  // ---- write C (bf16, original pair rows): C row = hi*4 + r within block
#pragma unroll
  for (int rb = 0; rb < 2; ++rb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int pr = cpair[rb][r];
      if (pr < 0) continue;
#pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        const int n = n0 + wave * 32 + nb * 16 + col;
        if (n < out_dim) c[(size_t)pr * out_dim + n] = f32_to_bf16(acc[rb][nb][r]);
      }
    }
}

// ------------------------------------------------------------------- host

torch::Tensor moe_gemm(
    c10::optional<torch::Tensor> wt_all,      // [E, in, out] bf16
    c10::optional<torch::Tensor> packed_all,  // [E, in, out/2] u8
    c10::optional<torch::Tensor> absmax_all,  // [E, in, out/64] bf16
    torch::Tensor x,                          // [T, in] bf16
    torch::Tensor sorted_pairs,               // [Rp] i32 (expert-sorted, MT-padded, -1 = pad)
    torch::Tensor tile_expert,                // [Rp/MT] i32 (-1 = pure padding tile)
    int64_t k_per_tok,
    int64_t n_rows) {                         // R = T * k_per_tok (C rows)
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(sorted_pairs.dtype() == torch::kInt32 && sorted_pairs.is_contiguous());
  TORCH_CHECK(tile_expert.dtype() == torch::kInt32 && tile_expert.is_contiguous());
  const bool nf4 = packed_all.has_value();
  int in_dim, out_dim;
  if (nf4) {
    TORCH_CHECK(packed_all->dtype() == torch::kUInt8 && packed_all->dim() == 3 && packed_all->is_contiguous());
    TORCH_CHECK(absmax_all.has_value() && absmax_all->dtype() == torch::kBFloat16);
    in_dim = packed_all->size(1);
    out_dim = packed_all->size(2) * 2;
  } else {
    TORCH_CHECK(wt_all.has_value() && wt_all->dtype() == torch::kBFloat16 && wt_all->dim() == 3);
    in_dim = wt_all->size(1);
    out_dim = wt_all->size(2);
  }
  TORCH_CHECK(x.size(1) == in_dim);
  // staging loads are 16B/32-value vectors: callers fall back to the dense
  // path for geometries outside these alignments (fused_moe.py)
  TORCH_CHECK(out_dim % MOE_NT == 0, "moe_gemm needs out_dim % 128 == 0");
  TORCH_CHECK(in_dim % MOE_KT == 0, "moe_gemm needs in_dim % 64 == 0");
  const int Rp = sorted_pairs.size(0);
  TORCH_CHECK(Rp % MOE_MT == 0 && tile_expert.size(0) == Rp / MOE_MT);

  auto c = torch::empty({n_rows, (long)out_dim}, x.options());
  dim3 grid(Rp / MOE_MT, (out_dim + MOE_NT - 1) / MOE_NT);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (nf4) {
    moe_gemm_kernel<true><<<grid, 256, 0, stream>>>(
        nullptr, packed_all->data_ptr<unsigned char>(),
        reinterpret_cast<const unsigned short*>(absmax_all->data_ptr()),
        reinterpret_cast<const unsigned short*>(x.data_ptr()),
        sorted_pairs.data_ptr<int>(), tile_expert.data_ptr<int>(),
        reinterpret_cast<unsigned short*>(c.data_ptr()), in_dim, out_dim, (int)k_per_tok);
  } else {
    moe_gemm_kernel<false><<<grid, 256, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(wt_all->data_ptr()), nullptr, nullptr,
        reinterpret_cast<const unsigned short*>(x.data_ptr()),
        sorted_pairs.data_ptr<int>(), tile_expert.data_ptr<int>(),
        reinterpret_cast<unsigned short*>(c.data_ptr()), in_dim, out_dim, (int)k_per_tok);
  }
  HIP_CHECK_LAST();
  return c;
}

torch::Tensor gemv_bf16_moe(
    torch::Tensor wt_all,    // [E, in, out] bf16
    torch::Tensor x,         // [B, in] f32
    torch::Tensor sel,       // [R] int32 (R = B * k)
    int64_t k_per_tok,
    torch::Tensor workspace,
    int64_t epilogue,
    int64_t splits_override) {
  TORCH_CHECK(wt_all.is_cuda() && wt_all.dtype() == torch::kBFloat16 && wt_all.dim() == 3);
  TORCH_CHECK(x.dtype() == torch::kFloat32 && x.dim() == 2);
  TORCH_CHECK(sel.dtype() == torch::kInt32 && sel.is_contiguous());
  const int in_dim = wt_all.size(1), out_dim = wt_all.size(2);
  const int rows = sel.size(0);
  TORCH_CHECK(x.size(1) == in_dim && rows == x.size(0) * k_per_tok);
  TORCH_CHECK(rows <= 16, "moe gemv supports batch*k <= 16");

  const long out_waves = (out_dim + GEMV_OUT_PER_WAVE - 1) / GEMV_OUT_PER_WAVE;
  long splits = splits_override > 0 ? splits_override : (768 + out_waves * rows - 1) / (out_waves * rows);
  long max_splits = (in_dim + 255) / 256;
  if (splits > max_splits) splits = max_splits;
  if (splits < 1) splits = 1;
  const int i_per_split = (in_dim + (int)splits - 1) / (int)splits;

  torch::Tensor partials;
  if (workspace.numel() >= (int64_t)splits * rows * out_dim) {
    partials = workspace;
  } else {
    partials = torch::empty({splits, (long)rows, (long)out_dim}, x.options());
  }
  dim3 grid(out_waves, splits, rows);
  auto stream = at::cuda::getCurrentCUDAStream();
  gemv_bf16_moe_kernel<<<grid, WAVE, 0, stream>>>(
      reinterpret_cast<const unsigned short*>(wt_all.data_ptr()), x.data_ptr<float>(),
      sel.data_ptr<int>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split,
      rows, (int)k_per_tok);
  HIP_CHECK_LAST();
  return launch_gemv_reduce(
      partials, c10::nullopt, c10::nullopt, (int)splits, rows, out_dim, (int)epilogue,
      x.options(), wt_all.options());
}

torch::Tensor gemv_nf4_moe(
    torch::Tensor packed_all,  // [E, in, out/2] u8
    torch::Tensor absmax_all,  // [E, in, out/64] bf16
    torch::Tensor x,           // [B, in] f32
    torch::Tensor sel,         // [R] int32
    int64_t k_per_tok,
    torch::Tensor workspace,
    int64_t epilogue,
    int64_t splits_override) {
  TORCH_CHECK(packed_all.is_cuda() && packed_all.dtype() == torch::kUInt8 && packed_all.dim() == 3);
  TORCH_CHECK(x.dtype() == torch::kFloat32 && x.dim() == 2);
  TORCH_CHECK(sel.dtype() == torch::kInt32 && sel.is_contiguous());
  const int in_dim = packed_all.size(1);
  const int out_dim = packed_all.size(2) * 2;
  const int rows = sel.size(0);
  TORCH_CHECK(x.size(1) == in_dim && rows == x.size(0) * k_per_tok);
  TORCH_CHECK(rows <= 16, "moe gemv supports batch*k <= 16");

  const long out_waves = (out_dim + NF4_OUT_PER_WAVE - 1) / NF4_OUT_PER_WAVE;
  long splits = splits_override > 0 ? splits_override : (1536 + out_waves * rows - 1) / (out_waves * rows);
  long max_splits = (in_dim + 31) / 32;
  if (splits > max_splits) splits = max_splits;
  if (splits < 1) splits = 1;
  const int i_per_split = (in_dim + (int)splits - 1) / (int)splits;

  torch::Tensor partials;
  if (workspace.numel() >= (int64_t)splits * rows * out_dim) {
    partials = workspace;
  } else {
    partials = torch::empty({splits, (long)rows, (long)out_dim}, x.options());
  }
  dim3 grid(out_waves, splits, rows);
  auto stream = at::cuda::getCurrentCUDAStream();
  gemv_nf4_moe_kernel<<<grid, WAVE, 0, stream>>>(
      packed_all.data_ptr<unsigned char>(),
      reinterpret_cast<const unsigned short*>(absmax_all.data_ptr()),
      x.data_ptr<float>(), sel.data_ptr<int>(), partials.data_ptr<float>(),
      in_dim, out_dim, i_per_split, rows, (int)k_per_tok);
  HIP_CHECK_LAST();
  return launch_gemv_reduce(
      partials, c10::nullopt, c10::nullopt, (int)splits, rows, out_dim, (int)epilogue,
      x.options(), absmax_all.options());
}
