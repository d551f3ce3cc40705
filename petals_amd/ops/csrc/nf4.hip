// NF4 4-bit blockwise quantization for CDNA4 (gfx950).
//
// Format (our own serving format; same 4.25 bits/param footprint as the
// reference's bitsandbytes LinearNF4 — blocksize 64, reference
// utils/convert_block.py:76-111, block_utils.py:46 — but with bf16 absmax
// instead of double-quantized int8 absmax: same size, better accuracy):
//   packed  uint8 [in, out/2]   two nibbles per byte (even elem = low nibble)
//   absmax  bf16  [in, out/64]  per-64-element scale along the OUT dim
//
// Weights stay TRANSPOSED [in, out] like the bf16 path, so the same
// x-broadcast GEMV structure applies; each lane owns 16 consecutive outputs
// (8 B packed per input row) and dequantizes in-register through an
// LDS-resident 16-entry NF4 LUT.

#include "common.h"
#include "gemv_reduce.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

__device__ __constant__ float NF4_LUT_C[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f, -0.39491748809814453f,
    -0.28444138169288635f, -0.18477343022823334f, -0.09105003625154495f, 0.0f,
    0.07958029955625534f, 0.16093020141124725f, 0.24611230194568634f,
    0.33791524171829224f, 0.4407098293304443f, 0.5626170039176941f,
    0.7229568362236023f, 1.0f};

// ------------------------------------------------------------- quantization

// one thread per 64-element block along OUT
__global__ void nf4_quantize_kernel(
    const unsigned short* __restrict__ w,  // [in, out] bf16
    unsigned char* __restrict__ packed,    // [in, out/2]
    unsigned short* __restrict__ absmax,   // [in, out/64] bf16
    long n_blocks,
    int out_dim) {
  const long blk = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (blk >= n_blocks) return;
  const long base = blk * 64;  // element offset (rows are multiples of 64 wide)
  float vals[64];
  float amax = 0.f;
#pragma unroll
  for (int j = 0; j < 64; ++j) {
    vals[j] = bf16_to_f32(w[base + j]);
    amax = fmaxf(amax, fabsf(vals[j]));
  }
  const float amax_bf = bf16_to_f32(f32_to_bf16(amax));  // store-rounded scale
  absmax[blk] = f32_to_bf16(amax);
  const float inv = amax_bf > 0.f ? 1.0f / amax_bf : 0.f;
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    unsigned char lo, hi;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const float v = vals[2 * j + half] * inv;
      // nearest NF4 level via linear scan (16 levels)
      int best = 0;
      float bd = fabsf(v - NF4_LUT_C[0]);
#pragma unroll
      for (int l = 1; l < 16; ++l) {
        const float d = fabsf(v - NF4_LUT_C[l]);
        if (d < bd) { bd = d; best = l; }
      }
      if (half == 0) lo = (unsigned char)best; else hi = (unsigned char)best;
    }
    packed[blk * 32 + j] = (unsigned char)(lo | (hi << 4));
  }
}

// --------------------------------------------------------------- dequantize

__global__ void nf4_dequant_kernel(
    const unsigned char* __restrict__ packed,
    const unsigned short* __restrict__ absmax,
    unsigned short* __restrict__ out,  // bf16
    long n_elems) {
  const long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 2;
  if (i >= n_elems) return;
  const unsigned char byte = packed[i >> 1];
  const float scale = bf16_to_f32(absmax[i >> 6]);
  out[i] = f32_to_bf16(NF4_LUT_C[byte & 0xF] * scale);
  if (i + 1 < n_elems) out[i + 1] = f32_to_bf16(NF4_LUT_C[byte >> 4] * scale);
}

// -------------------------------------------------------------- NF4 gemv

#define NF4_OUT_PER_WAVE 1024  // default: 64 lanes x 16 outputs (8 B packed / row)

// OPL = outputs per lane (16 or 8). OPL=8 doubles the workgroup count for the
// same split depth: the NF4 inner loop's dependent LDS gathers need more
// resident waves per SIMD to hide latency than the direct-load bf16 gemv.
// LUTBF: 4-byte bf16-pair LUT entries instead of 8-byte float2 — the LUT
// gathers read 8x the packed HBM bytes from LDS, so halving the entry width
// halves the dominant LDS-bandwidth term (2 extra VALU bit-ops per pair).
// XS: folded-RMSNorm input mode — x arrives as the raw bf16 hidden state and
// the kernel derives inv_rms from producer-side per-wg sum(h^2) partials
// (gemv_reduce sumsq_out), replacing the standalone norm kernel entirely (the
// norm WEIGHT is pre-folded into the quantized weight rows at load).
// LREP: bank-shifted replicas of the pair LUT. PMC measured 4.5 bank
// conflicts PER LDS INSTRUCTION on the single-copy LUT (64 random byte
// indices into 256 float2 entries = 2 KB over 32 banks-pairs); each replica
// is padded to 260 entries so copies land 8 banks apart, and lane l reads
// copy l & (LREP-1).
// UN: override of the weight-load unroll depth (0 = default 16, 8 for
// batch>4). The batch-1 kernel sits at 66 VGPRs with wait/busy ~12x
// (profiles/pmc_gemv_shapes_r2.txt) — deeper unroll trades registers for
// more HBM loads in flight.
template <int BATCH, int OPL, bool LUTBF, bool XS = false, int LREP = 1, int UN = 0>
__global__ __launch_bounds__(WAVE) void gemv_nf4_kernel(
    const unsigned char* __restrict__ packed,   // [in, out/2]
    const unsigned short* __restrict__ absmax,  // [in, out/64]
    const unsigned short* __restrict__ absmax_t,  // [out/64, in] or null — the
    // transposed copy turns the per-row 2-byte strided absmax gather into ONE
    // contiguous 32 B load per 16-row unroll block (the strided stream was a
    // latency-hiding bottleneck: 16 extra scattered loads in flight per block)
    const float* __restrict__ x,                // [BATCH, in] f32 (XS: unused)
    float* __restrict__ partials,               // [n_splits, BATCH, out]
    int in_dim,
    int out_dim,
    int i_per_split,
    const unsigned short* __restrict__ xb16 = nullptr,  // XS: [BATCH, in] bf16
    const float* __restrict__ xs_parts = nullptr,       // XS: [BATCH, n_parts] sum(h^2) partials
    int n_parts = 0,
    float eps = 0.f) {
  // PAIR LUT: one read dequantizes a whole packed byte (two elements),
  // halving LDS traffic vs per-nibble lookups; 256 entries.
  // (the earlier 16-float bank-replicated LUT was LDS-issue bound)
  __shared__ float2 lut2[LUTBF ? 1 : LREP][LUTBF ? 1 : 260];  // 260: replicas 8 banks apart
  __shared__ unsigned int lutp[LUTBF ? 256 : 1];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    if constexpr (LUTBF) {
      lutp[i] = ((unsigned int)f32_to_bf16(NF4_LUT_C[i >> 4]) << 16) |
                (unsigned int)f32_to_bf16(NF4_LUT_C[i & 0xF]);
    } else {
#pragma unroll
      for (int c = 0; c < LREP; ++c)
        lut2[c][i] = make_float2(NF4_LUT_C[i & 0xF], NF4_LUT_C[i >> 4]);
    }
  }
  __syncthreads();
  const int lrep = threadIdx.x & (LREP - 1);

  auto lut_pair = [&](unsigned int byte) -> float2 {
    if constexpr (LUTBF) {
      const unsigned int wp = lutp[byte];
      float2 r;
      r.x = __uint_as_float(wp << 16);          // low nibble (bf16 -> f32 bits)
      r.y = __uint_as_float(wp & 0xFFFF0000u);  // high nibble
      return r;
    } else {
      return lut2[lrep][byte];
    }
  };

  constexpr int WORDS = OPL / 8;  // u32 packed words per lane per row
  const int lane = threadIdx.x & (WAVE - 1);
  // inv_rms is a per-row CONSTANT, so it scales the ACCUMULATOR once at the
  // end instead of every x load: the sumsq-parts loads issue here (before the
  // weight stream starts) and their reduction happens after the main loop —
  // zero added latency on the critical path
  float pfp[XS ? 2 * BATCH : 1];
  if constexpr (XS) {
#pragma unroll
    for (int b = 0; b < BATCH; ++b) {
      pfp[2 * b] = lane < n_parts ? xs_parts[(size_t)b * n_parts + lane] : 0.f;
      pfp[2 * b + 1] = lane + WAVE < n_parts ? xs_parts[(size_t)b * n_parts + lane + WAVE] : 0.f;
    }
  }
#define NF4_L2(byte) lut_pair(byte)
  const int out0 = blockIdx.x * (WAVE * OPL) + lane * OPL;
  if (out0 >= out_dim) return;
  const int split = blockIdx.y;
  const int i_begin = split * i_per_split;
  const int i_end = min(i_begin + i_per_split, in_dim);
  const bool full = (out0 + OPL) <= out_dim;

  float acc[BATCH][OPL];
#pragma unroll
  for (int b = 0; b < BATCH; ++b)
#pragma unroll
    for (int v = 0; v < OPL; ++v) acc[b][v] = 0.f;

  if (full) {
    // batch > 4: halve the unroll depth — xs[BATCH][UNROLL] + acc[BATCH][OPL]
    // at full depth would blow the VGPR budget below 2 waves/SIMD
    constexpr int UNROLL = UN ? UN : ((BATCH > 4) ? 8 : 16);
    const int half_out = out_dim >> 1;
    const unsigned char* pp = packed + (size_t)i_begin * half_out + (out0 >> 1);
    const unsigned short* amt = absmax_t ? absmax_t + (size_t)(out0 >> 6) * in_dim : nullptr;
    int i = i_begin;
    for (; i + UNROLL <= i_end; i += UNROLL) {
      unsigned int pk[UNROLL][WORDS];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        if constexpr (WORDS == 2) {
          const uint2 p2v = *reinterpret_cast<const uint2*>(pp + (size_t)u * half_out);
          pk[u][0] = p2v.x;
          pk[u][1] = p2v.y;
        } else {
          pk[u][0] = *reinterpret_cast<const unsigned int*>(pp + (size_t)u * half_out);
        }
      }
      float am[UNROLL];
      if (amt) {
#pragma unroll
        for (int c8 = 0; c8 < UNROLL / 8; ++c8) {
          const short8 a8 = *reinterpret_cast<const short8*>(amt + i + c8 * 8);
#pragma unroll
          for (int e = 0; e < 8; ++e) am[c8 * 8 + e] = bf16_to_f32((unsigned short)a8[e]);
        }
      } else {
#pragma unroll
        for (int u = 0; u < UNROLL; ++u)
          am[u] = bf16_to_f32(absmax[(size_t)(i + u) * (out_dim >> 6) + (out0 >> 6)]);
      }
      // NB: scalar (SMEM) x loads share the lgkm counter with the LDS LUT
      // gathers, limiting ds_read pipelining — but both measured alternatives
      // (volatile VMEM: cache-bypassed broadcast re-reads; buffer-load
      // intrinsics: worse scheduling) LOST to this form end to end
      // (profiles/nf4_vmem_sweep.log), so the simple loads stay.
      float xs[BATCH][UNROLL];
#pragma unroll
      for (int b = 0; b < BATCH; ++b)
#pragma unroll
        for (int u = 0; u < UNROLL; ++u)
          xs[b][u] = XS ? bf16_to_f32(xb16[(size_t)b * in_dim + i + u])
                        : x[(size_t)b * in_dim + i + u];
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        float wf[OPL];
#pragma unroll
        for (int d = 0; d < WORDS; ++d)
#pragma unroll
          for (int p2 = 0; p2 < 4; ++p2) {
            const float2 w2 = NF4_L2((pk[u][d] >> (8 * p2)) & 0xFFu);
            wf[8 * d + 2 * p2] = w2.x;
            wf[8 * d + 2 * p2 + 1] = w2.y;
          }
#pragma unroll
        for (int b = 0; b < BATCH; ++b) {
          const float xa = xs[b][u] * am[u];
#pragma unroll
          for (int v = 0; v < OPL; ++v) acc[b][v] = fmaf(wf[v], xa, acc[b][v]);
        }
      }
      pp += (size_t)UNROLL * half_out;
    }
    for (; i < i_end; ++i) {
      const float am = bf16_to_f32(absmax[(size_t)i * (out_dim >> 6) + (out0 >> 6)]);
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = XS ? bf16_to_f32(xb16[(size_t)b * in_dim + i])
                            : x[(size_t)b * in_dim + i];
        const float xa = xv * am;
#pragma unroll
        for (int d = 0; d < WORDS; ++d) {
          const unsigned int wd =
              *reinterpret_cast<const unsigned int*>(packed + (size_t)i * half_out + (out0 >> 1) + 4 * d);
#pragma unroll
          for (int p2 = 0; p2 < 4; ++p2) {
            const float2 w2 = NF4_L2((wd >> (8 * p2)) & 0xFFu);
            acc[b][8 * d + 2 * p2] = fmaf(w2.x, xa, acc[b][8 * d + 2 * p2]);
            acc[b][8 * d + 2 * p2 + 1] = fmaf(w2.y, xa, acc[b][8 * d + 2 * p2 + 1]);
          }
        }
      }
    }
  } else {
    for (int i = i_begin; i < i_end; ++i) {
      const float am = bf16_to_f32(absmax[(size_t)i * (out_dim >> 6) + (out0 >> 6)]);
#pragma unroll
      for (int b = 0; b < BATCH; ++b) {
        const float xv = XS ? bf16_to_f32(xb16[(size_t)b * in_dim + i])
                            : x[(size_t)b * in_dim + i];
        const float xa = xv * am;
        // compile-time trip count with a guard: a runtime-bounded loop here
        // makes acc[][] dynamically indexed, forcing it onto the SCRATCH
        // stack for the whole kernel (observed at BATCH>=5: ~100x slowdown)
#pragma unroll
        for (int v = 0; v < OPL; ++v) {
          if (out0 + v < out_dim) {
            const unsigned char byte = packed[(size_t)i * (out_dim >> 1) + ((out0 + v) >> 1)];
            const float2 w2 = NF4_L2(byte);
            acc[b][v] = fmaf(((out0 + v) & 1) ? w2.y : w2.x, xa, acc[b][v]);
          }
        }
      }
    }
  }

  if constexpr (XS) {
#pragma unroll
    for (int b = 0; b < BATCH; ++b) {
      const float total = wave_reduce_sum(pfp[2 * b] + pfp[2 * b + 1]);
      const float inv = rsqrtf(total / in_dim + eps);
#pragma unroll
      for (int v = 0; v < OPL; ++v) acc[b][v] *= inv;
    }
  }
#pragma unroll
  for (int b = 0; b < BATCH; ++b) {
    float* dst = partials + ((size_t)split * BATCH + b) * out_dim + out0;
    if (full) {
#pragma unroll
      for (int q = 0; q < OPL / 4; ++q)
        reinterpret_cast<float4v*>(dst)[q] =
            float4v{acc[b][4 * q], acc[b][4 * q + 1], acc[b][4 * q + 2], acc[b][4 * q + 3]};
    } else {
#pragma unroll
      for (int v = 0; v < OPL; ++v)
        if (out0 + v < out_dim) dst[v] = acc[b][v];
    }
  }
}

// ------------------------------------------------------------------- host

std::vector<torch::Tensor> nf4_quantize(torch::Tensor w) {
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2);
  TORCH_CHECK(w.size(1) % 64 == 0, "out dim must be a multiple of 64");
  auto wc = w.contiguous();
  const long in_dim = wc.size(0), out_dim = wc.size(1);
  const long n_blocks = in_dim * out_dim / 64;
  auto packed = torch::empty({in_dim, out_dim / 2}, w.options().dtype(torch::kUInt8));
  auto absmax = torch::empty({in_dim, out_dim / 64}, w.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long blocks = (n_blocks + 255) / 256;
  nf4_quantize_kernel<<<blocks, 256, 0, stream>>>(
      reinterpret_cast<const unsigned short*>(wc.data_ptr()),
      packed.data_ptr<unsigned char>(),
      reinterpret_cast<unsigned short*>(absmax.data_ptr()),
      n_blocks, (int)out_dim);
  HIP_CHECK_LAST();
  return {packed, absmax};
}

torch::Tensor nf4_dequantize(torch::Tensor packed, torch::Tensor absmax) {
  TORCH_CHECK(packed.is_cuda() && packed.dtype() == torch::kUInt8 && packed.dim() == 2);
  const long in_dim = packed.size(0), half_out = packed.size(1);
  const long n = in_dim * half_out * 2;
  auto out = torch::empty({in_dim, half_out * 2}, absmax.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long blocks = (n / 2 + 255) / 256;
  nf4_dequant_kernel<<<blocks, 256, 0, stream>>>(
      packed.data_ptr<unsigned char>(),
      reinterpret_cast<const unsigned short*>(absmax.data_ptr()),
      reinterpret_cast<unsigned short*>(out.data_ptr()), n);
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor gemv_nf4(
    torch::Tensor packed,    // [in, out/2] u8
    torch::Tensor absmax,    // [in, out/64] bf16
    torch::Tensor x,         // [batch, in] f32 (folded-norm mode: bf16 raw hidden)
    torch::Tensor workspace,
    c10::optional<torch::Tensor> residual,
    int64_t epilogue,
    int64_t splits_override,
    c10::optional<torch::Tensor> bias,      // [out] bf16, added pre-activation
    c10::optional<torch::Tensor> absmax_t,  // [out/64, in] bf16 (transposed copy)
    c10::optional<torch::Tensor> x_parts,   // folded-norm: [batch, n_parts] sum(h^2) partials
    double fold_eps,
    c10::optional<torch::Tensor> sumsq_out) {  // [batch, out/64] f32: emit sum(y^2) per wg
  TORCH_CHECK(packed.is_cuda() && packed.dtype() == torch::kUInt8);
  const bool xs_mode = x_parts.has_value() && x_parts->defined() && x_parts->numel() > 0;
  if (xs_mode) {
    TORCH_CHECK(x.dtype() == torch::kBFloat16 && x.dim() == 2 && x.is_contiguous(),
                "folded-norm gemv takes the raw bf16 hidden state");
    TORCH_CHECK(x_parts->dtype() == torch::kFloat32 && x_parts->is_contiguous());
  } else {
    TORCH_CHECK(x.dtype() == torch::kFloat32 && x.dim() == 2);
  }
  const int in_dim = packed.size(0);
  const int out_dim = packed.size(1) * 2;
  const int batch = x.size(0);
  TORCH_CHECK(x.size(1) == in_dim && batch <= 8, "NF4 decode gemv supports batch <= 8");

  // outputs per lane: 8 doubles the grid vs 16 (latency hiding for the
  // LDS-gather dequant chain); PETALS_NF4_OPL overrides for A/B sweeps
  static const int opl_env = [] {
    const char* s = std::getenv("PETALS_NF4_OPL");
    return s ? std::atoi(s) : 0;
  }();
  const int opl = opl_env ? opl_env : 8;
  TORCH_CHECK(opl == 8 || opl == 16, "PETALS_NF4_OPL must be 8 or 16");
  TORCH_CHECK(!xs_mode || opl == 8, "folded-norm gemv is built for OPL=8");
  TORCH_CHECK(!xs_mode || x_parts->size(1) <= 2 * WAVE,
              "folded-norm gemv prefetches at most 128 sumsq partials");
  static const int lut_rep = [] {
    const char* s = std::getenv("PETALS_NF4_LUT_REP");
    return s ? std::atoi(s) : 1;
  }();
  static const int un_env = [] {
    const char* s = std::getenv("PETALS_NF4_UNROLL");
    return s ? std::atoi(s) : 0;
  }();
  static const bool lut_bf16 = [] {
    const char* s = std::getenv("PETALS_NF4_LUT");
    return s && s[0] == 'b';  // default f32 pairs (bf16-pair measured SLOWER:
    // the VALU unpack cost more than the halved LDS traffic bought,
    // profiles/nf4_lut_sweep.log — LDS bandwidth is not the binding limit)
  }();

  const int un32 = (un_env == 32 && batch <= 2 && opl == 8 && !lut_bf16) ? 32 : 0;
  const long out_waves = (out_dim + (long)WAVE * opl - 1) / ((long)WAVE * opl);
  // NF4 matrices are 4x smaller than bf16: allow chunks down to 64 input rows
  // so small projections still spread over the 256 CUs
  long splits = splits_override > 0 ? splits_override : (1536 + out_waves - 1) / out_waves;
  long max_splits = (in_dim + 31) / 32;
  if (splits > max_splits) splits = max_splits;
  if (splits < 1) splits = 1;
  // chunks must be UNROLL-aligned or the per-row fallback tail eats the
  // gain (an unaligned split count put up to a third of rows on the slow path)
  const long un_mask = un32 ? 31L : 15L;
  int i_per_split_aligned = (int)(((in_dim + splits - 1) / splits + un_mask) & ~un_mask);
  splits = (in_dim + i_per_split_aligned - 1) / i_per_split_aligned;

  torch::Tensor partials;
  auto f32opts = x.options().dtype(torch::kFloat32);
  if (workspace.numel() >= (int64_t)splits * batch * out_dim) {
    partials = workspace;
  } else {
    partials = torch::empty({(int64_t)splits, batch, out_dim}, f32opts);
  }
  const int i_per_split = i_per_split_aligned;
  dim3 grid(out_waves, splits);
  auto stream = at::cuda::getCurrentCUDAStream();

  const unsigned short* amt_p = nullptr;
  if (absmax_t.has_value() && absmax_t->defined() && absmax_t->numel() > 0) {
    TORCH_CHECK(absmax_t->is_contiguous() && absmax_t->size(0) == out_dim / 64
                && absmax_t->size(1) == in_dim);
    amt_p = reinterpret_cast<const unsigned short*>(absmax_t->data_ptr());
  }
#define LAUNCH_NF4(B, OPL, LB)                                                \
  if (lut_rep == 4 && OPL == 8 && !(LB))                                      \
    gemv_nf4_kernel<B, OPL, LB, false, 4><<<grid, WAVE, 0, stream>>>(         \
        packed.data_ptr<unsigned char>(),                                     \
        reinterpret_cast<const unsigned short*>(absmax.data_ptr()), amt_p,    \
        x.data_ptr<float>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split); \
  else                                                                        \
    gemv_nf4_kernel<B, OPL, LB><<<grid, WAVE, 0, stream>>>(                   \
        packed.data_ptr<unsigned char>(),                                     \
        reinterpret_cast<const unsigned short*>(absmax.data_ptr()), amt_p,    \
        x.data_ptr<float>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split)
#define LAUNCH_NF4_XS(B)                                                      \
  gemv_nf4_kernel<B, 8, false, true><<<grid, WAVE, 0, stream>>>(              \
      packed.data_ptr<unsigned char>(),                                       \
      reinterpret_cast<const unsigned short*>(absmax.data_ptr()), amt_p,      \
      nullptr, partials.data_ptr<float>(), in_dim, out_dim, i_per_split,      \
      reinterpret_cast<const unsigned short*>(x.data_ptr()),                  \
      x_parts->data_ptr<float>(), (int)x_parts->size(1), (float)fold_eps)
#define LAUNCH_NF4_B(OPL, LB)                                                 \
  switch (batch) {                                                            \
    case 1: LAUNCH_NF4(1, OPL, LB); break;                                    \
    case 2: LAUNCH_NF4(2, OPL, LB); break;                                    \
    case 3: LAUNCH_NF4(3, OPL, LB); break;                                    \
    case 4: LAUNCH_NF4(4, OPL, LB); break;                                    \
    case 5: LAUNCH_NF4(5, OPL, LB); break;                                    \
    case 6: LAUNCH_NF4(6, OPL, LB); break;                                    \
    case 7: LAUNCH_NF4(7, OPL, LB); break;                                    \
    case 8: LAUNCH_NF4(8, OPL, LB); break;                                    \
  }
  if (un32) {
    if (batch == 1)
      gemv_nf4_kernel<1, 8, false, false, 1, 32><<<grid, WAVE, 0, stream>>>(
          packed.data_ptr<unsigned char>(),
          reinterpret_cast<const unsigned short*>(absmax.data_ptr()), amt_p,
          x.data_ptr<float>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split);
    else
      gemv_nf4_kernel<2, 8, false, false, 1, 32><<<grid, WAVE, 0, stream>>>(
          packed.data_ptr<unsigned char>(),
          reinterpret_cast<const unsigned short*>(absmax.data_ptr()), amt_p,
          x.data_ptr<float>(), partials.data_ptr<float>(), in_dim, out_dim, i_per_split);
  } else if (xs_mode) {
    switch (batch) {
      case 1: LAUNCH_NF4_XS(1); break;
      case 2: LAUNCH_NF4_XS(2); break;
      case 3: LAUNCH_NF4_XS(3); break;
      case 4: LAUNCH_NF4_XS(4); break;
      case 5: LAUNCH_NF4_XS(5); break;
      case 6: LAUNCH_NF4_XS(6); break;
      case 7: LAUNCH_NF4_XS(7); break;
      case 8: LAUNCH_NF4_XS(8); break;
    }
  } else if (opl == 8) {
    if (lut_bf16) { LAUNCH_NF4_B(8, true) } else { LAUNCH_NF4_B(8, false) }
  } else {
    if (lut_bf16) { LAUNCH_NF4_B(16, true) } else { LAUNCH_NF4_B(16, false) }
  }
#undef LAUNCH_NF4_B
#undef LAUNCH_NF4_XS
#undef LAUNCH_NF4
  HIP_CHECK_LAST();

  if (epilogue < 0) {
    // RAW mode: the caller fuses its own reduce+epilogue (e.g. the qkv
    // rope+cache-write reduce); the view's shape carries the split count
    return partials.view(-1).narrow(0, 0, (int64_t)splits * batch * out_dim)
        .view({(int64_t)splits, (int64_t)batch, (int64_t)out_dim});
  }
  torch::Tensor y = launch_gemv_reduce(
      partials, residual, bias, (int)splits, batch, out_dim, (int)epilogue, f32opts,
      absmax.options(), sumsq_out);
  return y;
}
