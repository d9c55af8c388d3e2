// Fused BatchNorm + ReLU + clip kernels (NHWC layout on GPU).
//
// bn_stats: per-channel biased mean/var via a single pass over the NHWC
// tensor -- each block owns a channel-chunk and reduces sum / sumsq with
// f32 accumulation (channels are the fastest-varying dim in NHWC, so lanes
// read consecutive channels: fully coalesced).
// bn_act_fwd: one elementwise pass applying (x-mean)*invstd*gamma+beta,
// ReLU and the act_max clip.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

template <typename T>
struct alignas(16) BnPack8 { T v[8]; };

// Grid: (C + 63)/64 blocks in x, rows-chunks in y. Each block handles 64
// channels x kBlock/64-row slab. DETERMINISTIC: each block writes its
// per-channel partial to partial_sum[blockIdx.y * C + c] (no atomics; the
// intra-block combine iterates row-groups in a fixed order), and a second
// fixed-order pass reduces over blockIdx.y -- same-seed runs are
// bit-identical, which run-to-run atomic ordering was not.
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x,
                                float* __restrict__ partial_sum,
                                float* __restrict__ partial_sumsq,
                                int64_t rows, int C) {
  // channel/row thread split: each block owns a <=64-channel span and
  // packs kBlock/span row groups, so a partial span (C=65's lone tail
  // channel, MNv2's 16-32-channel stems) still uses every lane instead
  // of leaving one active lane per 64 as the critical path.
  __shared__ float ls[kBlock], lsq[kBlock];
  int cbase = blockIdx.x * 64;
  int cw = C - cbase;
  if (cw > 64) cw = 64;
  int c_l = (cw == 64) ? (threadIdx.x & 63) : (threadIdx.x % cw);
  int rgrp = (cw == 64) ? (threadIdx.x >> 6) : (threadIdx.x / cw);
  int ngrp = kBlock / cw;
  float acc_s = 0.0f, acc_q = 0.0f;
  if (rgrp < ngrp) {
    int c = cbase + c_l;
    int rstart = blockIdx.y * ngrp + rgrp;
    // 4 CONSECUTIVE rows per thread per iteration: the 4 in-flight loads
    // span one contiguous ~4*C*2B region (strided variants fetched 4
    // scattered 128-B lines each and ran 96% memory-wait)
    const int64_t st = (int64_t)gridDim.y * ngrp * 4;
    float s[4] = {}, sq[4] = {};
    int64_t r = (int64_t)rstart * 4;
    for (; r + 3 < rows; r += st) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float v = to_f32(x[(r + j) * C + c]);
        s[j] += v;
        sq[j] += v * v;
      }
    }
    for (; r < rows; ++r) {
      float v = to_f32(x[r * C + c]);
      s[0] += v;
      sq[0] += v * v;
    }
    acc_s = (s[0] + s[1]) + (s[2] + s[3]);
    acc_q = (sq[0] + sq[1]) + (sq[2] + sq[3]);
  }
  // slotted LDS (thread tid owns slot tid = rgrp*cw + c_l), combined in
  // FIXED row-group order by the channel threads
  ls[threadIdx.x] = acc_s;
  lsq[threadIdx.x] = acc_q;
  __syncthreads();
  if (threadIdx.x < cw) {
    float s = 0.0f, q = 0.0f;
    for (int gp = 0; gp < ngrp; ++gp) {
      s += ls[gp * cw + threadIdx.x];
      q += lsq[gp * cw + threadIdx.x];
    }
    int64_t o = (int64_t)blockIdx.y * C + cbase + threadIdx.x;
    partial_sum[o] = s;
    partial_sumsq[o] = q;
  }
}

// 8-channel-vector stats form (16-bit dtype, C % 8 == 0): 16-B row loads.
// Same deterministic partials contract as bn_stats_kernel.
template <typename T>
__global__ void bn_stats_vec_kernel(const T* __restrict__ x,
                                    float* __restrict__ partial_sum,
                                    float* __restrict__ partial_sumsq,
                                    int64_t rows, int C) {
  using V = BnPack8<T>;
  const int Cv = C >> 3;
  __shared__ float ls[kBlock * 8];
  __shared__ float lsq[kBlock * 8];
  int cvspan = Cv < 64 ? Cv : 64;
  int cbase_v = blockIdx.x * 64;
  int cw = Cv - cbase_v;
  if (cw > cvspan) cw = cvspan;
  int c_l = threadIdx.x % cw;
  int rgrp = threadIdx.x / cw;
  int ngrp = kBlock / cw;
  float acc_s[8] = {}, acc_q[8] = {};
  if (rgrp < ngrp) {
    int cv = cbase_v + c_l;
    int rstart = blockIdx.y * ngrp + rgrp;
    const int64_t st = (int64_t)gridDim.y * ngrp * 2;
    int64_t r = (int64_t)rstart * 2;
    for (; r + 1 < rows; r += st) {
      V a = ((const V*)x)[r * Cv + cv];
      V b = ((const V*)x)[(r + 1) * Cv + cv];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float va = to_f32(a.v[u]), vb = to_f32(b.v[u]);
        acc_s[u] += va + vb;
        acc_q[u] += va * va + vb * vb;
      }
    }
    for (; r < rows; ++r) {
      V a = ((const V*)x)[r * Cv + cv];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float va = to_f32(a.v[u]);
        acc_s[u] += va;
        acc_q[u] += va * va;
      }
    }
  }
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    ls[threadIdx.x * 8 + u] = acc_s[u];
    lsq[threadIdx.x * 8 + u] = acc_q[u];
  }
  __syncthreads();
  if (threadIdx.x < cw) {
    float s[8] = {}, q[8] = {};
    for (int gp = 0; gp < ngrp; ++gp) {
      int slot = (gp * cw + threadIdx.x) * 8;
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        s[u] += ls[slot + u];
        q[u] += lsq[slot + u];
      }
    }
    int64_t o = (int64_t)blockIdx.y * C + (cbase_v + threadIdx.x) * 8;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      partial_sum[o + u] = s[u];
      partial_sumsq[o + u] = q[u];
    }
  }
}

// Deterministic reduction of two [gy, C] partial arrays to [C].
// Parallel over rows as well as channels: 8 fixed row-strided streams per
// 32-channel chunk, combined in fixed stream order (the one-thread-per-
// channel serial loop was 4 waves crawling 1024 rows -- 7 ms/step on
// MobileNetV2). Every partition is a pure function of thread indices, so
// same-input runs are bit-identical.
__global__ void bn_partials_reduce_kernel(const float* __restrict__ pa,
                                          const float* __restrict__ pb,
                                          float* __restrict__ outa,
                                          float* __restrict__ outb, int gy,
                                          int C) {
  __shared__ float la[256], lb[256];
  int c = blockIdx.x * 32 + (threadIdx.x & 31);
  int rg = threadIdx.x >> 5;  // 8 row streams
  float a = 0.0f, b = 0.0f;
  if (c < C) {
    for (int r = rg; r < gy; r += 8) {
      a += pa[(int64_t)r * C + c];
      b += pb[(int64_t)r * C + c];
    }
  }
  la[threadIdx.x] = a;
  lb[threadIdx.x] = b;
  __syncthreads();
  if (threadIdx.x < 32 && c < C) {
    float sa = 0.0f, sb = 0.0f;
    for (int g = 0; g < 8; ++g) {
      sa += la[g * 32 + threadIdx.x];
      sb += lb[g * 32 + threadIdx.x];
    }
    outa[c] = sa;
    outb[c] = sb;
  }
}

template <typename T>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ invstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta, int64_t n,
                                  int C, int do_relu, float act_max) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float v = (to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (do_relu) v = fmaxf(v, 0.0f);
    if (act_max > 0.0f) v = fminf(v, act_max);
    y[i] = from_f32<T>(v);
  }
}

// 8-wide form (16-bit dtype, C % 8 == 0): 16-B loads/stores, channel
// params fetched per-vector (the scalar form issued one 2-B load per lane)
template <typename T>
__global__ void bn_act_fwd_vec_kernel(const T* __restrict__ x,
                                      T* __restrict__ y,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      int64_t n_vec, int Cv, int do_relu,
                                      float act_max) {
  using V = BnPack8<T>;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c0 = (int)(i % Cv) * 8;
    V xv = ((const V*)x)[i];
    V o;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      int c = c0 + u;
      float v = (to_f32(xv.v[u]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
      if (do_relu) v = fmaxf(v, 0.0f);
      if (act_max > 0.0f) v = fminf(v, act_max);
      o.v[u] = from_f32<T>(v);
    }
    ((V*)y)[i] = o;
  }
}

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = _Float16; };

// one tiny kernel replacing the eager mean/var/invstd/running-update chain
// (~8 launches per BN layer per step)
__global__ void bn_finalize_kernel(const float* __restrict__ partial_sum,
                                   const float* __restrict__ partial_sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, int gy,
                                   int C, float inv_n, float unbias,
                                   float momentum, float eps,
                                   int has_running) {
  // deterministic partials reduce folded in, parallel over 8 fixed row
  // streams per 32-channel chunk (see bn_partials_reduce_kernel)
  __shared__ float la[256], lb[256];
  int c = blockIdx.x * 32 + (threadIdx.x & 31);
  int rg = threadIdx.x >> 5;
  float a = 0.0f, b = 0.0f;
  if (c < C) {
    for (int r = rg; r < gy; r += 8) {
      a += partial_sum[(int64_t)r * C + c];
      b += partial_sumsq[(int64_t)r * C + c];
    }
  }
  la[threadIdx.x] = a;
  lb[threadIdx.x] = b;
  __syncthreads();
  if (threadIdx.x >= 32 || c >= C) return;
  float s = 0.0f, q = 0.0f;
  for (int g = 0; g < 8; ++g) {
    s += la[g * 32 + threadIdx.x];
    q += lb[g * 32 + threadIdx.x];
  }
  float m = s * inv_n;
  float v = fmaxf(q * inv_n - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(v + eps);
  if (has_running) {
    running_mean[c] = running_mean[c] * (1.0f - momentum) + momentum * m;
    running_var[c] =
        running_var[c] * (1.0f - momentum) + momentum * v * unbias;
  }
}

}  // namespace

// x: NHWC-contiguous 4-D (memory_format=channels_last, passed as NCHW logical)
// or 2-D [N, C] row-major. Returns (mean, biased var) as f32.
std::vector<torch::Tensor> bn_stats(torch::Tensor x) {
  int C;
  int64_t rows;
  const void* ptr = x.data_ptr();
  if (x.dim() == 4) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "bn_stats: expected channels_last");
    C = (int)x.size(1);
    rows = x.size(0) * x.size(2) * x.size(3);
  } else {
    TORCH_CHECK(x.is_contiguous());
    C = (int)x.size(1);
    rows = x.size(0);
  }
  auto opts = x.options().dtype(torch::kFloat32);
  bool vec16 = x.element_size() == 2 && (C & 7) == 0;
  int span = vec16 ? std::min(C / 8, 64) : std::min(C, 64);
  int gx = vec16 ? (C / 8 + 63) / 64 : (C + 63) / 64;
  int ngrp = kBlock / span;  // first block's packing
  int rper = vec16 ? 2 : 4;
  int gy = (int)std::min<int64_t>((rows + rper * ngrp - 1) / (rper * ngrp),
                                  std::max(1, 1024 / gx));
  auto partial_sum = torch::empty({gy, C}, opts);
  auto partial_sumsq = torch::empty({gy, C}, opts);
  auto sum = torch::empty({C}, opts);
  auto sumsq = torch::empty({C}, opts);
  auto stream = c10::hip::getCurrentHIPStream();
  NN_DISPATCH(x.scalar_type(),
                                  "bn_stats", [&] {
    using T = typename DevT<scalar_t>::type;
    if (vec16 && sizeof(T) == 2)
      hipLaunchKernelGGL((bn_stats_vec_kernel<T>), dim3(gx, gy), dim3(kBlock),
                         0, stream, (const T*)ptr,
                         partial_sum.data_ptr<float>(),
                         partial_sumsq.data_ptr<float>(), rows, C);
    else
      hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(gx, gy), dim3(kBlock), 0,
                         stream, (const T*)ptr, partial_sum.data_ptr<float>(),
                         partial_sumsq.data_ptr<float>(), rows, C);
  });
  hipLaunchKernelGGL(bn_partials_reduce_kernel,
                     dim3((C + 31) / 32), dim3(256), 0, stream,
                     partial_sum.data_ptr<float>(),
                     partial_sumsq.data_ptr<float>(), sum.data_ptr<float>(),
                     sumsq.data_ptr<float>(), gy, C);
  HIP_CHECK_LAST();
  auto mean = sum / (double)rows;
  auto var = sumsq / (double)rows - mean * mean;
  return {mean, var.clamp_min(0)};
}

torch::Tensor bn_act_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor invstd, torch::Tensor gamma,
                         torch::Tensor beta, bool relu, double act_max) {
  int C;
  if (x.dim() == 4) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
    C = (int)x.size(1);
  } else {
    TORCH_CHECK(x.is_contiguous());
    C = (int)x.size(1);
  }
  auto y = torch::empty_like(x);
  int64_t n = x.numel();
  NN_DISPATCH(x.scalar_type(),
                                  "bn_act_fwd", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    if (sizeof(T) == 2 && (C & 7) == 0) {
      int64_t n_vec = n / 8;
      int blocks = (int)std::min<int64_t>((n_vec + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((bn_act_fwd_vec_kernel<T>), dim3(blocks),
                         dim3(kBlock), 0, stream, (const T*)x.data_ptr(),
                         (T*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                         beta.data_ptr<float>(), n_vec, C / 8, relu ? 1 : 0,
                         (float)act_max);
    } else {
      int blocks = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 8192);
      hipLaunchKernelGGL((bn_act_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0,
                         stream, (const T*)x.data_ptr(),
                         (T*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                         beta.data_ptr<float>(), n, C, relu ? 1 : 0,
                         (float)act_max);
    }
  });
  HIP_CHECK_LAST();
  return y;
}

// ---------------------------------------------------------------------------
// Fused BN+act backward: (1) per-channel reduce of (g*mask) and
// (g*mask*xhat), (2) elementwise gx. The activation mask (ReLU/clip) is
// re-derived from the saved OUTPUT y, folding the act backward into the BN
// backward (the reference pays separate eager kernels for each).
// ---------------------------------------------------------------------------

namespace {

template <typename T>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ g, const T* __restrict__ x, const T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ partial_g, float* __restrict__ partial_gx,
    int64_t rows, int C, int do_relu, float act_max) {
  __shared__ float ls[kBlock], lsq[kBlock];  // see bn_stats_kernel
  int cbase = blockIdx.x * 64;
  int cw = C - cbase;
  if (cw > 64) cw = 64;
  int c_l = (cw == 64) ? (threadIdx.x & 63) : (threadIdx.x % cw);
  int rgrp = (cw == 64) ? (threadIdx.x >> 6) : (threadIdx.x / cw);
  int ngrp = kBlock / cw;
  float acc_g = 0.0f, acc_gx = 0.0f;
  if (rgrp < ngrp) {
    int c = cbase + c_l;
    int rstart = blockIdx.y * ngrp + rgrp;
    float m = mean[c], is = invstd[c];
    // 4 consecutive rows per iteration: see bn_stats_kernel
    const int64_t st = (int64_t)gridDim.y * ngrp * 4;
    float s_g[4] = {}, s_gx[4] = {};
    auto body = [&](int64_t r, int j) {
      int64_t i = r * C + c;
      float yv = to_f32(y[i]);
      float mask = 1.0f;
      if (do_relu && yv <= 0.0f) mask = 0.0f;
      if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
      float gv = to_f32(g[i]) * mask;
      s_g[j] += gv;
      s_gx[j] += gv * (to_f32(x[i]) - m) * is;
    };
    int64_t r = (int64_t)rstart * 4;
    for (; r + 3 < rows; r += st) {
#pragma unroll
      for (int j = 0; j < 4; ++j) body(r + j, j);
    }
    for (; r < rows; ++r) body(r, 0);
    acc_g = (s_g[0] + s_g[1]) + (s_g[2] + s_g[3]);
    acc_gx = (s_gx[0] + s_gx[1]) + (s_gx[2] + s_gx[3]);
  }
  // slotted LDS + fixed-order combine (see bn_stats_kernel)
  ls[threadIdx.x] = acc_g;
  lsq[threadIdx.x] = acc_gx;
  __syncthreads();
  if (threadIdx.x < cw) {
    float sg = 0.0f, sgx = 0.0f;
    for (int gp = 0; gp < ngrp; ++gp) {
      sg += ls[gp * cw + threadIdx.x];
      sgx += lsq[gp * cw + threadIdx.x];
    }
    int64_t o = (int64_t)blockIdx.y * C + cbase + threadIdx.x;
    partial_g[o] = sg;
    partial_gx[o] = sgx;
  }
}

// 8-channel-vector backward reduce (16-bit dtype, C % 8 == 0)
template <typename T>
__global__ void bn_act_bwd_reduce_vec_kernel(
    const T* __restrict__ g, const T* __restrict__ x, const T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ partial_g, float* __restrict__ partial_gx,
    int64_t rows, int C, int do_relu, float act_max) {
  using V = BnPack8<T>;
  const int Cv = C >> 3;
  __shared__ float ls[kBlock * 8];
  __shared__ float lsq[kBlock * 8];
  int cvspan = Cv < 64 ? Cv : 64;
  int cbase_v = blockIdx.x * 64;
  int cw = Cv - cbase_v;
  if (cw > cvspan) cw = cvspan;
  int c_l = threadIdx.x % cw;
  int rgrp = threadIdx.x / cw;
  int ngrp = kBlock / cw;
  float acc_g[8] = {}, acc_gx[8] = {};
  if (rgrp < ngrp) {
    int cv = cbase_v + c_l;
    int c0 = cv * 8;
    float m[8], is[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      m[u] = mean[c0 + u];
      is[u] = invstd[c0 + u];
    }
    // 2 CONSECUTIVE rows in flight per iteration: 6 independent loads
    // hide more latency than the single-row chain
    int rstart = blockIdx.y * ngrp + rgrp;
    const int64_t st = (int64_t)gridDim.y * ngrp * 2;
    int64_t r = (int64_t)rstart * 2;
    for (; r + 1 < rows; r += st) {
      int64_t o0 = r * Cv + cv;
      int64_t o1 = (r + 1) * Cv + cv;
      V ga = ((const V*)g)[o0], gb = ((const V*)g)[o1];
      V xa = ((const V*)x)[o0], xb = ((const V*)x)[o1];
      V ya = ((const V*)y)[o0], yb = ((const V*)y)[o1];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float yv = to_f32(ya.v[u]);
        float mask = 1.0f;
        if (do_relu && yv <= 0.0f) mask = 0.0f;
        if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
        float gv = to_f32(ga.v[u]) * mask;
        acc_g[u] += gv;
        acc_gx[u] += gv * (to_f32(xa.v[u]) - m[u]) * is[u];
        float yv2 = to_f32(yb.v[u]);
        float mask2 = 1.0f;
        if (do_relu && yv2 <= 0.0f) mask2 = 0.0f;
        if (act_max > 0.0f && yv2 >= act_max) mask2 = 0.0f;
        float gv2 = to_f32(gb.v[u]) * mask2;
        acc_g[u] += gv2;
        acc_gx[u] += gv2 * (to_f32(xb.v[u]) - m[u]) * is[u];
      }
    }
    for (; r < rows; ++r) {
      int64_t o = r * Cv + cv;
      V gv8 = ((const V*)g)[o];
      V xv8 = ((const V*)x)[o];
      V yv8 = ((const V*)y)[o];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float yv = to_f32(yv8.v[u]);
        float mask = 1.0f;
        if (do_relu && yv <= 0.0f) mask = 0.0f;
        if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
        float gv = to_f32(gv8.v[u]) * mask;
        acc_g[u] += gv;
        acc_gx[u] += gv * (to_f32(xv8.v[u]) - m[u]) * is[u];
      }
    }
  }
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    ls[threadIdx.x * 8 + u] = acc_g[u];
    lsq[threadIdx.x * 8 + u] = acc_gx[u];
  }
  __syncthreads();
  if (threadIdx.x < cw) {
    float s[8] = {}, q[8] = {};
    for (int gp = 0; gp < ngrp; ++gp) {
      int slot = (gp * cw + threadIdx.x) * 8;
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        s[u] += ls[slot + u];
        q[u] += lsq[slot + u];
      }
    }
    int64_t o = (int64_t)blockIdx.y * C + (cbase_v + threadIdx.x) * 8;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      partial_g[o + u] = s[u];
      partial_gx[o + u] = q[u];
    }
  }
}

template <typename T, bool TRAIN>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ g, const T* __restrict__ x, const T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ sum_g,
    const float* __restrict__ sum_gx, T* __restrict__ gx, int64_t n, int C,
    float inv_count, int do_relu, float act_max) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    float yv = to_f32(y[i]);
    float mask = 1.0f;
    if (do_relu && yv <= 0.0f) mask = 0.0f;
    if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
    float gv = to_f32(g[i]) * mask;
    float gi = gamma[c] * invstd[c];
    if (TRAIN) {
      float xhat = (to_f32(x[i]) - mean[c]) * invstd[c];
      gv = gi * (gv - sum_g[c] * inv_count - xhat * sum_gx[c] * inv_count);
    } else {
      gv = gv * gi;
    }
    gx[i] = from_f32<T>(gv);
  }
}

template <typename T, bool TRAIN>
__global__ void bn_act_bwd_apply_vec_kernel(
    const T* __restrict__ g, const T* __restrict__ x, const T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ gamma, const float* __restrict__ sum_g,
    const float* __restrict__ sum_gx, T* __restrict__ gx, int64_t n_vec,
    int Cv, float inv_count, int do_relu, float act_max) {
  using V = BnPack8<T>;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c0 = (int)(i % Cv) * 8;
    V gv8 = ((const V*)g)[i];
    V yv8 = ((const V*)y)[i];
    V xv8;
    if (TRAIN) xv8 = ((const V*)x)[i];
    V o;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      int c = c0 + u;
      float yv = to_f32(yv8.v[u]);
      float mask = 1.0f;
      if (do_relu && yv <= 0.0f) mask = 0.0f;
      if (act_max > 0.0f && yv >= act_max) mask = 0.0f;
      float gv = to_f32(gv8.v[u]) * mask;
      float gi = gamma[c] * invstd[c];
      if (TRAIN) {
        float xhat = (to_f32(xv8.v[u]) - mean[c]) * invstd[c];
        gv = gi * (gv - sum_g[c] * inv_count - xhat * sum_gx[c] * inv_count);
      } else {
        gv = gv * gi;
      }
      o.v[u] = from_f32<T>(gv);
    }
    ((V*)gx)[i] = o;
  }
}

}  // namespace

std::vector<torch::Tensor> bn_act_bwd(torch::Tensor g, torch::Tensor x,
                                      torch::Tensor y, torch::Tensor mean,
                                      torch::Tensor invstd, torch::Tensor gamma,
                                      bool training, bool relu,
                                      double act_max) {
  int C;
  int64_t rows;
  if (x.dim() == 4) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(g.is_contiguous(at::MemoryFormat::ChannelsLast));
    C = (int)x.size(1);
    rows = x.size(0) * x.size(2) * x.size(3);
  } else {
    TORCH_CHECK(x.is_contiguous() && g.is_contiguous());
    C = (int)x.size(1);
    rows = x.size(0);
  }
  auto opts = x.options().dtype(torch::kFloat32);
  auto sum_g = torch::empty({C}, opts);
  auto sum_gx = torch::empty({C}, opts);
  auto gx = torch::empty_like(g);
  auto stream = c10::hip::getCurrentHIPStream();
  bool vec16 = x.element_size() == 2 && (C & 7) == 0;
  int span = vec16 ? std::min(C / 8, 64) : std::min(C, 64);
  int gx_blocks = vec16 ? (C / 8 + 63) / 64 : (C + 63) / 64;
  int ngrp = kBlock / span;  // first block's packing
  int rper = vec16 ? 2 : 4;
  int gy = (int)std::min<int64_t>((rows + rper * ngrp - 1) / (rper * ngrp),
                                  std::max(1, 1024 / gx_blocks));
  auto partial_g = torch::empty({gy, C}, opts);
  auto partial_gx = torch::empty({gy, C}, opts);
  int64_t n = x.numel();
  int eblocks = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(x.scalar_type(), "bn_act_bwd", [&] {
    using T = typename DevT<scalar_t>::type;
    if (vec16 && sizeof(T) == 2)
      hipLaunchKernelGGL((bn_act_bwd_reduce_vec_kernel<T>),
                         dim3(gx_blocks, gy), dim3(kBlock), 0, stream,
                         (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                         (const T*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         partial_g.data_ptr<float>(),
                         partial_gx.data_ptr<float>(),
                         rows, C, relu ? 1 : 0, (float)act_max);
    else
      hipLaunchKernelGGL((bn_act_bwd_reduce_kernel<T>), dim3(gx_blocks, gy),
                         dim3(kBlock), 0, stream, (const T*)g.data_ptr(),
                         (const T*)x.data_ptr(), (const T*)y.data_ptr(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         partial_g.data_ptr<float>(),
                         partial_gx.data_ptr<float>(),
                         rows, C, relu ? 1 : 0, (float)act_max);
    hipLaunchKernelGGL(bn_partials_reduce_kernel,
                       dim3((C + 31) / 32), dim3(256), 0,
                       stream, partial_g.data_ptr<float>(),
                       partial_gx.data_ptr<float>(), sum_g.data_ptr<float>(),
                       sum_gx.data_ptr<float>(), gy, C);
    bool vec = sizeof(T) == 2 && (C & 7) == 0;
    int vblocks = (int)std::min<int64_t>((n / 8 + kBlock - 1) / kBlock, 8192);
    if (training) {
      if (vec)
        hipLaunchKernelGGL((bn_act_bwd_apply_vec_kernel<T, true>),
                           dim3(vblocks), dim3(kBlock), 0, stream,
                           (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                           (const T*)y.data_ptr(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gx.data_ptr<float>(),
                           (T*)gx.data_ptr(), n / 8, C / 8,
                           1.0f / (float)rows, relu ? 1 : 0, (float)act_max);
      else
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<T, true>), dim3(eblocks),
                           dim3(kBlock), 0, stream, (const T*)g.data_ptr(),
                           (const T*)x.data_ptr(), (const T*)y.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), sum_g.data_ptr<float>(),
                           sum_gx.data_ptr<float>(), (T*)gx.data_ptr(), n, C,
                           1.0f / (float)rows, relu ? 1 : 0, (float)act_max);
    } else {
      if (vec)
        hipLaunchKernelGGL((bn_act_bwd_apply_vec_kernel<T, false>),
                           dim3(vblocks), dim3(kBlock), 0, stream,
                           (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                           (const T*)y.data_ptr(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gx.data_ptr<float>(),
                           (T*)gx.data_ptr(), n / 8, C / 8,
                           1.0f / (float)rows, relu ? 1 : 0, (float)act_max);
      else
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<T, false>), dim3(eblocks),
                           dim3(kBlock), 0, stream, (const T*)g.data_ptr(),
                           (const T*)x.data_ptr(), (const T*)y.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), sum_g.data_ptr<float>(),
                           sum_gx.data_ptr<float>(), (T*)gx.data_ptr(), n, C,
                           1.0f / (float)rows, relu ? 1 : 0, (float)act_max);
    }
  });
  HIP_CHECK_LAST();
  // g_gamma = sum_gx (already masked * xhat), g_beta = sum_g
  return {gx, sum_gx, sum_g};
}


// Split backward for SyncBN: reduce -> (RCCL all-reduce in Python) -> apply.
std::vector<torch::Tensor> bn_act_bwd_reduce(torch::Tensor g, torch::Tensor x,
                                             torch::Tensor y, torch::Tensor mean,
                                             torch::Tensor invstd, bool relu,
                                             double act_max) {
  int C;
  int64_t rows;
  if (x.dim() == 4) {
    C = (int)x.size(1);
    rows = x.size(0) * x.size(2) * x.size(3);
  } else {
    C = (int)x.size(1);
    rows = x.size(0);
  }
  auto opts = x.options().dtype(torch::kFloat32);
  auto sum_g = torch::empty({C}, opts);
  auto sum_gx = torch::empty({C}, opts);
  bool vec16 = x.element_size() == 2 && (C & 7) == 0;
  int span = vec16 ? std::min(C / 8, 64) : std::min(C, 64);
  int gx_blocks = vec16 ? (C / 8 + 63) / 64 : (C + 63) / 64;
  int ngrp = kBlock / span;  // first block's packing
  int rper = vec16 ? 2 : 4;
  int gy = (int)std::min<int64_t>((rows + rper * ngrp - 1) / (rper * ngrp),
                                  std::max(1, 1024 / gx_blocks));
  auto partial_g = torch::empty({gy, C}, opts);
  auto partial_gx = torch::empty({gy, C}, opts);
  auto stream = c10::hip::getCurrentHIPStream();
  NN_DISPATCH(x.scalar_type(), "bn_act_bwd_reduce", [&] {
    using T = typename DevT<scalar_t>::type;
    if (vec16 && sizeof(T) == 2)
      hipLaunchKernelGGL((bn_act_bwd_reduce_vec_kernel<T>),
                         dim3(gx_blocks, gy), dim3(kBlock), 0, stream,
                         (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                         (const T*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         partial_g.data_ptr<float>(),
                         partial_gx.data_ptr<float>(), rows, C, relu ? 1 : 0,
                         (float)act_max);
    else
      hipLaunchKernelGGL((bn_act_bwd_reduce_kernel<T>), dim3(gx_blocks, gy),
                         dim3(kBlock), 0, stream,
                         (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                         (const T*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), partial_g.data_ptr<float>(),
                         partial_gx.data_ptr<float>(), rows, C, relu ? 1 : 0,
                         (float)act_max);
  });
  hipLaunchKernelGGL(bn_partials_reduce_kernel,
                     dim3((C + 31) / 32), dim3(256), 0, stream,
                     partial_g.data_ptr<float>(), partial_gx.data_ptr<float>(),
                     sum_g.data_ptr<float>(), sum_gx.data_ptr<float>(), gy, C);
  HIP_CHECK_LAST();
  return {sum_g, sum_gx};
}

torch::Tensor bn_act_bwd_apply(torch::Tensor g, torch::Tensor x,
                               torch::Tensor y, torch::Tensor mean,
                               torch::Tensor invstd, torch::Tensor gamma,
                               torch::Tensor sum_g, torch::Tensor sum_gx,
                               double count, bool training, bool relu,
                               double act_max) {
  int C = (int)x.size(1);
  auto gx = torch::empty_like(g);
  int64_t n = x.numel();
  int eblocks = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 8192);
  NN_DISPATCH(x.scalar_type(), "bn_act_bwd_apply", [&] {
    using T = typename DevT<scalar_t>::type;
    auto stream = c10::hip::getCurrentHIPStream();
    bool vec = sizeof(T) == 2 && (C & 7) == 0;
    int vblocks = (int)std::min<int64_t>((n / 8 + kBlock - 1) / kBlock, 8192);
    if (training) {
      if (vec)
        hipLaunchKernelGGL((bn_act_bwd_apply_vec_kernel<T, true>),
                           dim3(vblocks), dim3(kBlock), 0, stream,
                           (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                           (const T*)y.data_ptr(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gx.data_ptr<float>(),
                           (T*)gx.data_ptr(), n / 8, C / 8,
                           1.0f / (float)count, relu ? 1 : 0, (float)act_max);
      else
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<T, true>), dim3(eblocks),
                           dim3(kBlock), 0, stream, (const T*)g.data_ptr(),
                           (const T*)x.data_ptr(), (const T*)y.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), sum_g.data_ptr<float>(),
                           sum_gx.data_ptr<float>(), (T*)gx.data_ptr(), n, C,
                           1.0f / (float)count, relu ? 1 : 0, (float)act_max);
    } else {
      if (vec)
        hipLaunchKernelGGL((bn_act_bwd_apply_vec_kernel<T, false>),
                           dim3(vblocks), dim3(kBlock), 0, stream,
                           (const T*)g.data_ptr(), (const T*)x.data_ptr(),
                           (const T*)y.data_ptr(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gx.data_ptr<float>(),
                           (T*)gx.data_ptr(), n / 8, C / 8,
                           1.0f / (float)count, relu ? 1 : 0, (float)act_max);
      else
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<T, false>), dim3(eblocks),
                           dim3(kBlock), 0, stream, (const T*)g.data_ptr(),
                           (const T*)x.data_ptr(), (const T*)y.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), sum_g.data_ptr<float>(),
                           sum_gx.data_ptr<float>(), (T*)gx.data_ptr(), n, C,
                           1.0f / (float)count, relu ? 1 : 0, (float)act_max);
    }
  });
  HIP_CHECK_LAST();
  return gx;
}


// batch stats + finalize in one shot: raw sum/sumsq pass, then one tiny
// kernel computes (mean, invstd) and updates the (f32) running buffers
// in-place -- replacing the ~8 eager launches of the Python chain.
// Single-rank path only; SyncBN keeps the raw (mean, E[x^2]) route so the
// all-reduce can sit between stats and normalization.
std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor x,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var,
                                             double momentum, double eps) {
  int C;
  int64_t rows;
  const void* ptr = x.data_ptr();
  if (x.dim() == 4) {
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
                "bn_stats_finalize: expected channels_last");
    C = (int)x.size(1);
    rows = x.size(0) * x.size(2) * x.size(3);
  } else {
    TORCH_CHECK(x.is_contiguous());
    C = (int)x.size(1);
    rows = x.size(0);
  }
  bool has_running = running_mean.numel() > 0;
  TORCH_CHECK(!has_running || (running_mean.scalar_type() == torch::kFloat32 &&
                               running_var.scalar_type() == torch::kFloat32),
              "bn_stats_finalize: running stats must be f32");
  auto opts = x.options().dtype(torch::kFloat32);
  bool vec16 = x.element_size() == 2 && (C & 7) == 0;
  int span = vec16 ? std::min(C / 8, 64) : std::min(C, 64);
  int gx = vec16 ? (C / 8 + 63) / 64 : (C + 63) / 64;
  int ngrp = kBlock / span;  // first block's packing
  int rper = vec16 ? 2 : 4;
  int gy = (int)std::min<int64_t>((rows + rper * ngrp - 1) / (rper * ngrp),
                                  std::max(1, 1024 / gx));
  auto partial_sum = torch::empty({gy, C}, opts);
  auto partial_sumsq = torch::empty({gy, C}, opts);
  auto stream = c10::hip::getCurrentHIPStream();
  NN_DISPATCH(x.scalar_type(), "bn_stats_finalize", [&] {
    using T = typename DevT<scalar_t>::type;
    if (vec16 && sizeof(T) == 2)
      hipLaunchKernelGGL((bn_stats_vec_kernel<T>), dim3(gx, gy), dim3(kBlock),
                         0, stream, (const T*)ptr,
                         partial_sum.data_ptr<float>(),
                         partial_sumsq.data_ptr<float>(), rows, C);
    else
      hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(gx, gy), dim3(kBlock), 0,
                         stream, (const T*)ptr, partial_sum.data_ptr<float>(),
                         partial_sumsq.data_ptr<float>(), rows, C);
  });
  float n = (float)rows;
  float unbias = n / std::max(n - 1.0f, 1.0f);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 31) / 32), dim3(256), 0,
                     stream, partial_sum.data_ptr<float>(),
                     partial_sumsq.data_ptr<float>(),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     has_running ? running_mean.data_ptr<float>() : nullptr,
                     has_running ? running_var.data_ptr<float>() : nullptr, gy,
                     C, 1.0f / n, unbias, (float)momentum, (float)eps,
                     has_running ? 1 : 0);
  HIP_CHECK_LAST();
  return {mean, invstd};
}
