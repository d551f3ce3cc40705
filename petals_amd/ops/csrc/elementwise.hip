// Elementwise / normalization kernels for CDNA4 (gfx950): RMSNorm, LayerNorm,
// RoPE + KV-cache write, SwiGLU. All bf16 IO vectorized as short8 (G13 of the
// CDNA4 guide: hipcc does not auto-vectorize bf16 loads).

#include "common.h"
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

// --------------------------------------------------------------- rms_norm

// one workgroup per row; row held in registers (short8 chunks). T=1024 for
// single-row decode calls (one wg is the whole launch — fewer serial chunk
// phases shorten the latency chain), T=256 for many-row prefill.
template <bool OUT_F32, int T>
__global__ void rms_norm_kernel(
    const unsigned short* __restrict__ x,  // [rows, dim] bf16
    const unsigned short* __restrict__ w,  // [dim] bf16
    void* __restrict__ y,                  // [rows, dim] bf16 or f32
    int dim,
    float eps) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned short* xr = x + (size_t)row * dim;

  float sumsq = 0.f;
  const int chunk = T * 8;
  for (int base = 0; base < dim; base += chunk) {
    const int idx = base + tid * 8;
    if (idx + 8 <= dim) {
      const short8 v = *reinterpret_cast<const short8*>(xr + idx);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf16_to_f32((unsigned short)v[j]);
        sumsq = fmaf(f, f, sumsq);
      }
    } else {
      for (int j = idx; j < dim; ++j) {
        const float f = bf16_to_f32(xr[j]);
        sumsq = fmaf(f, f, sumsq);
      }
    }
  }
  __shared__ float red[T / WAVE];
  sumsq = wave_reduce_sum(sumsq);
  if ((tid & (WAVE - 1)) == 0) red[tid / WAVE] = sumsq;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < T / WAVE; ++i) total += red[i];
  const float inv = rsqrtf(total / dim + eps);

  for (int base = 0; base < dim; base += chunk) {
    const int idx = base + tid * 8;
    if (idx + 8 <= dim) {
      const short8 v = *reinterpret_cast<const short8*>(xr + idx);
      const short8 wv = *reinterpret_cast<const short8*>(w + idx);
      if (OUT_F32) {
        float* yr = reinterpret_cast<float*>(y) + (size_t)row * dim + idx;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          yr[j] = bf16_to_f32((unsigned short)v[j]) * inv * bf16_to_f32((unsigned short)wv[j]);
      } else {
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          out[j] = (short)f32_to_bf16(bf16_to_f32((unsigned short)v[j]) * inv * bf16_to_f32((unsigned short)wv[j]));
        *reinterpret_cast<short8*>(reinterpret_cast<unsigned short*>(y) + (size_t)row * dim + idx) = out;
      }
    } else {
      for (int j = idx; j < dim; ++j) {
        const float f = bf16_to_f32(xr[j]) * inv * bf16_to_f32(w[j]);
        if (OUT_F32)
          reinterpret_cast<float*>(y)[(size_t)row * dim + j] = f;
        else
          reinterpret_cast<unsigned short*>(y)[(size_t)row * dim + j] = f32_to_bf16(f);
      }
    }
  }
}

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "rms_norm expects bf16 CUDA tensor");
  auto shape = x.sizes().vec();
  const int dim = shape.back();
  auto x2 = x.contiguous().view({-1, dim});
  const int rows = x2.size(0);
  auto y = torch::empty_like(x2);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (rows <= 4)
    rms_norm_kernel<false, 1024><<<rows, 1024, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(x2.data_ptr()),
        reinterpret_cast<const unsigned short*>(w.contiguous().data_ptr()),
        y.data_ptr(), dim, (float)eps);
  else
    rms_norm_kernel<false, 256><<<rows, 256, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(x2.data_ptr()),
        reinterpret_cast<const unsigned short*>(w.contiguous().data_ptr()),
        y.data_ptr(), dim, (float)eps);
  HIP_CHECK_LAST();
  return y.view(shape);
}

torch::Tensor rms_norm_f32out(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  auto shape = x.sizes().vec();
  const int dim = shape.back();
  auto x2 = x.contiguous().view({-1, dim});
  const int rows = x2.size(0);
  auto y = torch::empty({rows, dim}, x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (rows <= 4)
    rms_norm_kernel<true, 1024><<<rows, 1024, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(x2.data_ptr()),
        reinterpret_cast<const unsigned short*>(w.contiguous().data_ptr()),
        y.data_ptr(), dim, (float)eps);
  else
    rms_norm_kernel<true, 256><<<rows, 256, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(x2.data_ptr()),
        reinterpret_cast<const unsigned short*>(w.contiguous().data_ptr()),
        y.data_ptr(), dim, (float)eps);
  HIP_CHECK_LAST();
  return y.view(shape);
}

// ------------------------------------------------------------- layer_norm

// one workgroup (256 threads) per row, two-pass mean/variance in registers;
// BLOOM/Falcon blocks use LayerNorm with bias where Llama uses RMSNorm
// (reference models/bloom/block.py wraps HF BloomBlock whose norms run in
// torch; here the norm is one fused kernel feeding the gemv in f32).
template <bool OUT_F32, int T>
__global__ void layer_norm_kernel(
    const unsigned short* __restrict__ x,  // [rows, dim] bf16
    const unsigned short* __restrict__ w,  // [dim] bf16
    const unsigned short* __restrict__ bias,  // [dim] bf16
    void* __restrict__ y,                  // [rows, dim] bf16 or f32
    int dim,
    float eps) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned short* xr = x + (size_t)row * dim;

  float sum = 0.f, sumsq = 0.f;
  const int chunk = T * 8;
  for (int base = 0; base < dim; base += chunk) {
    const int idx = base + tid * 8;
    if (idx + 8 <= dim) {
      const short8 v = *reinterpret_cast<const short8*>(xr + idx);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf16_to_f32((unsigned short)v[j]);
        sum += f;
        sumsq = fmaf(f, f, sumsq);
      }
    } else {
      for (int j = idx; j < dim; ++j) {
        const float f = bf16_to_f32(xr[j]);
        sum += f;
        sumsq = fmaf(f, f, sumsq);
      }
    }
  }
  __shared__ float red_s[T / WAVE];
  __shared__ float red_q[T / WAVE];
  sum = wave_reduce_sum(sum);
  sumsq = wave_reduce_sum(sumsq);
  if ((tid & (WAVE - 1)) == 0) {
    red_s[tid / WAVE] = sum;
    red_q[tid / WAVE] = sumsq;
  }
  __syncthreads();
  float tot_s = 0.f, tot_q = 0.f;
#pragma unroll
  for (int i = 0; i < T / WAVE; ++i) {
    tot_s += red_s[i];
    tot_q += red_q[i];
  }
  const float mean = tot_s / dim;
  const float var = tot_q / dim - mean * mean;
  const float inv = rsqrtf(var + eps);

  for (int base = 0; base < dim; base += chunk) {
    const int idx = base + tid * 8;
    if (idx + 8 <= dim) {
      const short8 v = *reinterpret_cast<const short8*>(xr + idx);
      const short8 wv = *reinterpret_cast<const short8*>(w + idx);
      const short8 bv = *reinterpret_cast<const short8*>(bias + idx);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = (bf16_to_f32((unsigned short)v[j]) - mean) * inv *
                            bf16_to_f32((unsigned short)wv[j]) +
                        bf16_to_f32((unsigned short)bv[j]);
        if (OUT_F32)
          reinterpret_cast<float*>(y)[(size_t)row * dim + idx + j] = f;
        else
          reinterpret_cast<unsigned short*>(y)[(size_t)row * dim + idx + j] = f32_to_bf16(f);
      }
    } else {
      for (int j = idx; j < dim; ++j) {
        const float f = (bf16_to_f32(xr[j]) - mean) * inv * bf16_to_f32(w[j]) + bf16_to_f32(bias[j]);
        if (OUT_F32)
          reinterpret_cast<float*>(y)[(size_t)row * dim + j] = f;
        else
          reinterpret_cast<unsigned short*>(y)[(size_t)row * dim + j] = f32_to_bf16(f);
      }
    }
  }
}

static torch::Tensor layer_norm_impl(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps, bool f32out) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "layer_norm expects bf16 CUDA tensor");
  TORCH_CHECK(w.dtype() == torch::kBFloat16 && b.dtype() == torch::kBFloat16);
  auto shape = x.sizes().vec();
  const int dim = shape.back();
  auto x2 = x.contiguous().view({-1, dim});
  const int rows = x2.size(0);
  auto y = f32out ? torch::empty({rows, dim}, x.options().dtype(torch::kFloat32)) : torch::empty_like(x2);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto* xp = reinterpret_cast<const unsigned short*>(x2.data_ptr());
  auto* wp = reinterpret_cast<const unsigned short*>(w.contiguous().data_ptr());
  auto* bp = reinterpret_cast<const unsigned short*>(b.contiguous().data_ptr());
  if (rows <= 4) {
    if (f32out)
      layer_norm_kernel<true, 1024><<<rows, 1024, 0, stream>>>(xp, wp, bp, y.data_ptr(), dim, (float)eps);
    else
      layer_norm_kernel<false, 1024><<<rows, 1024, 0, stream>>>(xp, wp, bp, y.data_ptr(), dim, (float)eps);
  } else {
    if (f32out)
      layer_norm_kernel<true, 256><<<rows, 256, 0, stream>>>(xp, wp, bp, y.data_ptr(), dim, (float)eps);
    else
      layer_norm_kernel<false, 256><<<rows, 256, 0, stream>>>(xp, wp, bp, y.data_ptr(), dim, (float)eps);
  }
  HIP_CHECK_LAST();
  return f32out ? y : y.view(shape);
}

torch::Tensor layer_norm(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps) {
  auto shape = x.sizes().vec();
  auto y = layer_norm_impl(x, w, b, eps, false);
  return y.view(shape);
}

torch::Tensor layer_norm_f32out(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps) {
  return layer_norm_impl(x, w, b, eps, true);
}

// --------------------------------------------------------------- warm_spin

// one tiny workgroup that occupies the device queue for ~us microseconds
// (wall_clock64 ticks at a fixed ~100 MHz, clock-independent). Used by the
// scheduler's opt-in keep-warm path: per-token serving leaves ~10 ms host
// gaps in which aggressive power management parks the clocks, making the
// next step run several times slower; a queue that never drains avoids the
// idle state at negligible power (s_sleep in the loop).
__global__ void warm_spin_kernel(long long ticks) {
  const long long start = wall_clock64();
  while (wall_clock64() - start < ticks) __builtin_amdgcn_s_sleep(32);
}

void warm_spin(int64_t microseconds) {
  const long long us = std::min<int64_t>(std::max<int64_t>(microseconds, 1), 2000);
  auto stream = at::cuda::getCurrentCUDAStream();
  warm_spin_kernel<<<1, 64, 0, stream>>>(us * 100);  // 100 MHz wall clock
  HIP_CHECK_LAST();
}

// ------------------------------------------------------------------ swiglu

__global__ void swiglu_kernel(
    const unsigned short* __restrict__ gate,
    const unsigned short* __restrict__ up,
    unsigned short* __restrict__ y,
    long n) {
  const long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (i + 8 <= n) {
    const short8 g8 = *reinterpret_cast<const short8*>(gate + i);
    const short8 u8 = *reinterpret_cast<const short8*>(up + i);
    short8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float g = bf16_to_f32((unsigned short)g8[j]);
      const float u = bf16_to_f32((unsigned short)u8[j]);
      o8[j] = (short)f32_to_bf16(g / (1.f + __expf(-g)) * u);
    }
    *reinterpret_cast<short8*>(y + i) = o8;
  } else {
    for (long j = i; j < n; ++j) {
      const float g = bf16_to_f32(gate[j]);
      const float u = bf16_to_f32(up[j]);
      y[j] = f32_to_bf16(g / (1.f + __expf(-g)) * u);
    }
  }
}

torch::Tensor swiglu(torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.dtype() == torch::kBFloat16);
  auto g = gate.contiguous(), u = up.contiguous();
  auto y = torch::empty_like(g);
  const long n = g.numel();
  const long blocks = std::min((n / 8 + 255) / 256 + 1, (long)4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  swiglu_kernel<<<blocks, 256, 0, stream>>>(
      reinterpret_cast<const unsigned short*>(g.data_ptr()),
      reinterpret_cast<const unsigned short*>(u.data_ptr()),
      reinterpret_cast<unsigned short*>(y.data_ptr()), n);
  HIP_CHECK_LAST();
  return y;
}

// ----------------------------------------------- rope (batched, bf16 q/k)

// q,k: [b, heads, len, hd] bf16; cos/sin: [max_pos, hd] f32; positions [b, len]
__global__ void rope_kernel(
    unsigned short* __restrict__ q,
    unsigned short* __restrict__ k,
    const float* __restrict__ cos_t,
    const float* __restrict__ sin_t,
    const long* __restrict__ pos,  // [b, len]
    int b, int qh, int kh, int len, int hd) {
  const int half = hd / 2;
  const long total = (long)b * (qh + kh) * len * half;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long t = idx;
    const int d = t % half; t /= half;
    const int l = t % len; t /= len;
    const int h = t % (qh + kh); t /= (qh + kh);
    const int bi = t;
    unsigned short* base = (h < qh)
        ? q + (((size_t)bi * qh + h) * len + l) * hd
        : k + (((size_t)bi * kh + (h - qh)) * len + l) * hd;
    const long p = pos[(size_t)bi * len + l];
    const float c = cos_t[(size_t)p * hd + d];
    const float s = sin_t[(size_t)p * hd + d];
    const float x1 = bf16_to_f32(base[d]);
    const float x2 = bf16_to_f32(base[d + half]);
    base[d] = f32_to_bf16(x1 * c - x2 * s);
    base[d + half] = f32_to_bf16(x2 * c + x1 * s);
  }
}

// returns (q, k) rotated copies
std::vector<torch::Tensor> apply_rope(
    torch::Tensor q, torch::Tensor k, torch::Tensor cos_t, torch::Tensor sin_t, torch::Tensor pos) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && k.dim() == 4);
  TORCH_CHECK(q.dtype() == torch::kBFloat16 && k.dtype() == torch::kBFloat16);
  auto qc = q.contiguous().clone();
  auto kc = k.contiguous().clone();
  auto cf = cos_t.to(torch::kFloat32).contiguous();
  auto sf = sin_t.to(torch::kFloat32).contiguous();
  auto pc = pos.to(torch::kInt64).contiguous();
  const int b = qc.size(0), qh = qc.size(1), len = qc.size(2), hd = qc.size(3);
  const int kh = kc.size(1);
  TORCH_CHECK(pc.dim() == 2 && pc.size(0) == b && pc.size(1) == len, "positions must be [b, len]");
  const long total = (long)b * (qh + kh) * len * (hd / 2);
  const long blocks = std::min((total + 255) / 256 + 1, (long)4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  rope_kernel<<<blocks, 256, 0, stream>>>(
      reinterpret_cast<unsigned short*>(qc.data_ptr()),
      reinterpret_cast<unsigned short*>(kc.data_ptr()),
      cf.data_ptr<float>(), sf.data_ptr<float>(), pc.data_ptr<long>(),
      b, qh, kh, len, hd);
  HIP_CHECK_LAST();
  return {qc, kc};
}

// -------------------------- fused decode rope + kv-cache write (fast path)

// qkv: [b, qh*hd + 2*kh*hd] f32 (one token per row). Rotates q in place,
// rotates k and writes k/v into the caches (bf16) at row *pos_ptr.
__global__ void rope_cache_write_kernel(
    float* __restrict__ qkv,
    const float* __restrict__ cos_t,
    const float* __restrict__ sin_t,
    const int* __restrict__ pos_ptr,
    unsigned short* __restrict__ k_cache,  // [bcap, kh, lmax, hd]
    unsigned short* __restrict__ v_cache,
    int b, int qh, int kh, int lmax, int hd) {
  const int half = hd / 2;
  const int p = *pos_ptr;
  const float* crow = cos_t + (size_t)p * hd;
  const float* srow = sin_t + (size_t)p * hd;
  const int row_elems = qh * hd + 2 * kh * hd;
  // rotate q+k pairs: b * (qh+kh) * half work items; copy v: b * kh * hd
  const long rot_total = (long)b * (qh + kh) * half;
  const long v_total = (long)b * kh * hd;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < rot_total + v_total;
       idx += (long)gridDim.x * blockDim.x) {
    if (idx < rot_total) {
      long t = idx;
      const int d = t % half; t /= half;
      const int h = t % (qh + kh); t /= (qh + kh);
      const int bi = t;
      const float c = crow[d], s = srow[d];
      if (h < qh) {
        float* base = qkv + (size_t)bi * row_elems + h * hd;
        const float x1 = base[d], x2 = base[d + half];
        base[d] = x1 * c - x2 * s;
        base[d + half] = x2 * c + x1 * s;
      } else {
        const int kh_i = h - qh;
        float* base = qkv + (size_t)bi * row_elems + qh * hd + kh_i * hd;
        const float x1 = base[d], x2 = base[d + half];
        unsigned short* krow = k_cache + (((size_t)bi * kh + kh_i) * lmax + p) * hd;
        krow[d] = f32_to_bf16(x1 * c - x2 * s);
        krow[d + half] = f32_to_bf16(x2 * c + x1 * s);
      }
    } else {
      long t = idx - rot_total;
      const int d = t % hd; t /= hd;
      const int kh_i = t % kh; t /= kh;
      const int bi = t;
      const float v = qkv[(size_t)bi * row_elems + qh * hd + kh * hd + kh_i * hd + d];
      v_cache[(((size_t)bi * kh + kh_i) * lmax + p) * hd + d] = f32_to_bf16(v);
    }
  }
}

// ---------------- fused qkv split-K reduce + rope + kv-cache write ----------
// Consumes the qkv gemv's RAW partials ([splits, b, qh*hd+2*kh*hd]) and in ONE
// kernel: sums the splits (same 64-outputs x G split-group structure as
// gemv_reduce), rotates q/k pairs by the device-resident position, writes
// rotated q to a fresh [b, qh*hd] f32 tensor and k/v (bf16) into the caches.
// Replaces reduce + rope_cache_write (or reduce + kv_cache_write for ALiBi
// families) — one fewer kernel per block on the decode path.
#define QKV_RED_GROUPS 16  // 16 split groups x 64 outputs (1024-thread wgs):
// halves each thread's serial split chain vs 8 — measured +0.9-1.0 tok/s on
// every alternating pair (profiles/qrr_groups_ab.log); PETALS_QRR_GROUPS=8
// selects the old width

template <bool ROPE, int G_T = QKV_RED_GROUPS>
__global__ __launch_bounds__(64 * G_T) void qkv_rope_reduce_kernel(
    const float* __restrict__ partials,  // [splits, b, row_elems]
    const float* __restrict__ cos_t,     // [max_pos, hd]
    const float* __restrict__ sin_t,
    const int* __restrict__ pos_ptr,
    const unsigned short* __restrict__ bias,  // [row_elems] bf16 or null
    float* __restrict__ q_out,             // [b, qh*hd] f32 (rotated)
    unsigned short* __restrict__ k_cache,  // [bcap, kh, lmax, hd]
    unsigned short* __restrict__ v_cache,
    int n_splits, int b, int qh, int kh, int lmax, int hd) {
  constexpr int G = G_T;
  __shared__ float acc_a[G][64];
  __shared__ float acc_b[G][64];
  const int half = hd >> 1;
  const int row_elems = qh * hd + 2 * kh * hd;
  const int total = b * row_elems;
  const int ol = threadIdx.x & 63;
  const int sg = threadIdx.x >> 6;
  const int idx = blockIdx.x * 64 + ol;
  const bool live = idx < total;
  const int bi = live ? idx / row_elems : 0;
  const int o = live ? idx - bi * row_elems : 0;
  const size_t stride = (size_t)b * row_elems;
  const size_t row = (size_t)bi * row_elems;

  // does this output need its rotation partner's sum too?
  const bool in_q = o < qh * hd;
  const bool in_k = !in_q && o < qh * hd + kh * hd;
  const int d = in_q ? (o % hd) : (in_k ? ((o - qh * hd) % hd) : 0);
  const bool rotate = ROPE && (in_q || in_k);
  // partner offset: +half for d<half, -half otherwise
  const int partner = rotate ? (d < half ? o + half : o - half) : o;

  float a = 0.f, p2 = 0.f;
  if (live) {
    for (int s = sg; s < n_splits; s += G) a += partials[(size_t)s * stride + row + o];
    if (rotate)
      for (int s = sg; s < n_splits; s += G) p2 += partials[(size_t)s * stride + row + partner];
  }
  acc_a[sg][ol] = a;
  acc_b[sg][ol] = p2;
  __syncthreads();
  if (sg != 0 || !live) return;
  float sum = 0.f, psum = 0.f;
#pragma unroll
  for (int k = 0; k < G; ++k) {
    sum += acc_a[k][ol];
    psum += acc_b[k][ol];
  }
  if (bias) {
    sum += bf16_to_f32(bias[o]);
    if (rotate) psum += bf16_to_f32(bias[partner]);
  }

  const int p = *pos_ptr;
  float val = sum;
  if (rotate) {
    const float c = cos_t[(size_t)p * hd + (d < half ? d : d - half)];
    const float s = sin_t[(size_t)p * hd + (d < half ? d : d - half)];
    // pair (x1, x2) at (d, d+half): out_d = x1*c - x2*s; out_{d+half} = x2*c + x1*s
    val = (d < half) ? (sum * c - psum * s) : (sum * c + psum * s);
  }
  if (in_q) {
    q_out[(size_t)bi * qh * hd + o] = val;
  } else if (in_k) {
    const int kh_i = (o - qh * hd) / hd;
    k_cache[(((size_t)bi * kh + kh_i) * lmax + p) * hd + d] = f32_to_bf16(val);
  } else {
    const int t = o - qh * hd - kh * hd;
    const int kh_i = t / hd;
    const int dv = t - kh_i * hd;
    v_cache[(((size_t)bi * kh + kh_i) * lmax + p) * hd + dv] = f32_to_bf16(val);
  }
}

torch::Tensor qkv_rope_reduce(
    torch::Tensor partials,  // [splits, b, row_elems] f32 (RAW gemv output)
    c10::optional<torch::Tensor> cos_t,
    c10::optional<torch::Tensor> sin_t,
    torch::Tensor pos,       // device int32 [1]
    torch::Tensor k_cache,   // [bcap, kh, lmax, hd] bf16
    torch::Tensor v_cache,
    int64_t qh, int64_t kh, bool rope,
    c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(partials.is_cuda() && partials.dtype() == torch::kFloat32 && partials.dim() == 3);
  const int n_splits = partials.size(0);
  const int b = partials.size(1);
  const int kh_i = k_cache.size(1), lmax = k_cache.size(2), hd = k_cache.size(3);
  TORCH_CHECK(kh_i == kh);
  TORCH_CHECK(partials.size(2) == qh * hd + 2 * kh * hd, "qkv width mismatch");
  auto q_out = torch::empty({(int64_t)b, qh * hd}, partials.options());
  const int total = b * (int)partials.size(2);
  const int blocks = (total + 63) / 64;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* cp = nullptr;
  const float* sp = nullptr;
  if (rope) {
    TORCH_CHECK(cos_t.has_value() && sin_t.has_value(), "rope needs cos/sin tables");
    cp = cos_t->data_ptr<float>();
    sp = sin_t->data_ptr<float>();
  }
  const unsigned short* bp = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    TORCH_CHECK(bias->is_contiguous() && bias->dtype() == torch::kBFloat16);
    TORCH_CHECK(bias->numel() == partials.size(2));
    bp = reinterpret_cast<const unsigned short*>(bias->data_ptr());
  }
  static const int qrr_g = [] {
    const char* e = std::getenv("PETALS_QRR_GROUPS");
    return e ? std::atoi(e) : QKV_RED_GROUPS;
  }();
#define LAUNCH_QRR_G(R, G)                                                     \
  qkv_rope_reduce_kernel<R, G><<<blocks, 64 * G, 0, stream>>>(                 \
      partials.data_ptr<float>(), cp, sp, pos.data_ptr<int>(), bp,             \
      q_out.data_ptr<float>(),                                                 \
      reinterpret_cast<unsigned short*>(k_cache.data_ptr()),                   \
      reinterpret_cast<unsigned short*>(v_cache.data_ptr()),                   \
      n_splits, b, (int)qh, (int)kh, lmax, hd)
  if (qrr_g == 8) {
    if (rope) LAUNCH_QRR_G(true, 8); else LAUNCH_QRR_G(false, 8);
  } else {
    if (rope) LAUNCH_QRR_G(true, 16); else LAUNCH_QRR_G(false, 16);
  }
#undef LAUNCH_QRR_G
  HIP_CHECK_LAST();
  return q_out;
}

// kv cache write WITHOUT rope for ALiBi families (BLOOM): same [q|k|v] fused
// layout as rope_cache_write, k and v copied bf16 into the caches at *pos.
__global__ void kv_cache_write_kernel(
    const float* __restrict__ qkv,
    const int* __restrict__ pos_ptr,
    unsigned short* __restrict__ k_cache,  // [bcap, kh, lmax, hd]
    unsigned short* __restrict__ v_cache,
    int b, int qh, int kh, int lmax, int hd) {
  const int p = *pos_ptr;
  const int row_elems = qh * hd + 2 * kh * hd;
  const long total = (long)b * kh * hd;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < 2 * total;
       idx += (long)gridDim.x * blockDim.x) {
    const bool is_v = idx >= total;
    long t = is_v ? idx - total : idx;
    const int d = t % hd; t /= hd;
    const int kh_i = t % kh; t /= kh;
    const int bi = t;
    const float val = qkv[(size_t)bi * row_elems + qh * hd + (is_v ? kh * hd : 0) + kh_i * hd + d];
    unsigned short* cache = is_v ? v_cache : k_cache;
    cache[(((size_t)bi * kh + kh_i) * lmax + p) * hd + d] = f32_to_bf16(val);
  }
}

void kv_cache_write(
    torch::Tensor qkv, torch::Tensor pos,
    torch::Tensor k_cache, torch::Tensor v_cache, int64_t qh, int64_t kh) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kFloat32 && qkv.dim() == 2);
  TORCH_CHECK(k_cache.dtype() == torch::kBFloat16 && k_cache.dim() == 4);
  const int b = qkv.size(0);
  const int hd = k_cache.size(3), lmax = k_cache.size(2);
  const long total = 2L * b * kh * hd;
  const long blocks = std::min((total + 255) / 256 + 1, (long)1024);
  auto stream = at::cuda::getCurrentCUDAStream();
  kv_cache_write_kernel<<<blocks, 256, 0, stream>>>(
      qkv.data_ptr<float>(), pos.data_ptr<int>(),
      reinterpret_cast<unsigned short*>(k_cache.data_ptr()),
      reinterpret_cast<unsigned short*>(v_cache.data_ptr()),
      b, (int)qh, (int)kh, lmax, hd);
  HIP_CHECK_LAST();
}

void rope_cache_write(
    torch::Tensor qkv, torch::Tensor cos_t, torch::Tensor sin_t, torch::Tensor pos,
    torch::Tensor k_cache, torch::Tensor v_cache, int64_t qh, int64_t kh) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kFloat32 && qkv.dim() == 2);
  TORCH_CHECK(k_cache.dtype() == torch::kBFloat16 && k_cache.dim() == 4);
  const int b = qkv.size(0);
  const int hd = k_cache.size(3), lmax = k_cache.size(2);
  const long total = (long)b * (qh + kh) * (hd / 2) + (long)b * kh * hd;
  const long blocks = std::min((total + 255) / 256 + 1, (long)1024);
  auto stream = at::cuda::getCurrentCUDAStream();
  rope_cache_write_kernel<<<blocks, 256, 0, stream>>>(
      qkv.data_ptr<float>(), cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
      pos.data_ptr<int>(),
      reinterpret_cast<unsigned short*>(k_cache.data_ptr()),
      reinterpret_cast<unsigned short*>(v_cache.data_ptr()),
      b, (int)qh, (int)kh, lmax, hd);
  HIP_CHECK_LAST();
}

// ---------------- row sum-of-squares (folded-RMSNorm first-block feeder) ----
// One workgroup per row: parts layout matches gemv_reduce's sumsq_out
// ([rows, 1] here — the consuming gemv sums n_parts entries either way).

__global__ void sumsq_rows_kernel(
    const unsigned short* __restrict__ x,  // [rows, dim] bf16
    float* __restrict__ parts,             // [rows, 1]
    int dim) {
  const int row = blockIdx.x;
  const unsigned short* xr = x + (size_t)row * dim;
  float s = 0.f;
  for (int i = threadIdx.x * 8; i + 8 <= dim; i += blockDim.x * 8) {
    const short8 v = *reinterpret_cast<const short8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf16_to_f32((unsigned short)v[j]);
      s = fmaf(f, f, s);
    }
  }
  for (int i = (dim & ~7) + threadIdx.x; i < dim; i += blockDim.x) {
    const float f = bf16_to_f32(xr[i]);
    s = fmaf(f, f, s);
  }
  __shared__ float red[1024 / WAVE];
  s = wave_reduce_sum(s);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int k = 0; k < (int)(blockDim.x / WAVE); ++k) t += red[k];
    parts[row] = t;
  }
}

torch::Tensor sumsq_rows(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2 && x.is_contiguous());
  const int rows = x.size(0), dim = x.size(1);
  auto parts = torch::empty({rows, 1}, x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  sumsq_rows_kernel<<<rows, 1024, 0, stream>>>(
      reinterpret_cast<const unsigned short*>(x.data_ptr()), parts.data_ptr<float>(), dim);
  HIP_CHECK_LAST();
  return parts;
}
