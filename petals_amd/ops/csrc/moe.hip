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

static __device__ __constant__ float MOE_NF4_LUT[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f, -0.39491748809814453f,
    -0.28444138169288635f, -0.18477343022823334f, -0.09105003625154495f, 0.0f,
    0.07958029955625534f, 0.16093020141124725f, 0.24611230194568634f,
    0.33791524171829224f, 0.4407098293304443f, 0.5626170039176941f,
    0.7229568362236023f, 1.0f};

// ------------------------------------------------------------ bf16 indexed

__global__ void gemv_bf16_moe_kernel(
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
    const int nv = out_dim - out0;
    for (int i = i_begin; i < i_end; ++i) {
      const float xv = xr[i];
      for (int v = 0; v < nv; ++v)
        acc[v] = fmaf(bf16_to_f32(wt[(size_t)i * out_dim + out0 + v]), xv, acc[v]);
    }
  }

  float* dst = partials + ((size_t)split * rows + row) * out_dim + out0;
  if (full) {
    float4v* d4 = reinterpret_cast<float4v*>(dst);
    d4[0] = float4v{acc[0], acc[1], acc[2], acc[3]};
    d4[1] = float4v{acc[4], acc[5], acc[6], acc[7]};
  } else {
    for (int v = 0; v < out_dim - out0; ++v) dst[v] = acc[v];
  }
}

// ------------------------------------------------------------- nf4 indexed

__global__ void gemv_nf4_moe_kernel(
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
      for (int v = 0; v < out_dim - out0; ++v) {
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
    for (int v = 0; v < out_dim - out0; ++v) dst[v] = acc[v];
  }
}

// ------------------------------------------------------------------- host

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
