// K-means kernel suite for the cluster-routing index (IVF).
//
// Replaces the reference's 9-kernel Metal suite
// (reference: pkg/gpu/metal/kmeans_kernels_darwin.metal:71-370 — distance
// matrix, argmin assign, zero/accumulate/finalize centroids, drift,
// single-point reassign, k-means++ distances) with CDNA4-native designs:
//
//  - k_kmeans_assign<NV>: fused distance + argmin, one wave per point.
//    The point's dims live in registers (NV x short8v per lane); each
//    centroid row is streamed through the wave (32 B/lane coalesced,
//    L2-resident: k x d bf16 is ~0.5 MB at k=223) and reduced with
//    v_dot2c_f32_bf16 + 6-step shuffle. argmin uses d2 = |c|^2 - 2 x.c
//    (|x|^2 is constant per point) and reports full squared distance.
//  - k_kmeans_accum: hierarchical accumulate — per-point vector add into
//    fp32 sums via global atomics (random cluster mix -> low contention),
//    counts once per point.
//  - k_kmeans_finalize: sums/counts -> new centroids (empty clusters keep
//    their position), plus per-cluster squared drift (wave per cluster).
//  - k_kmeanspp_update: d2 = min(d2, dist2(x, c)) for one new seed.
//  - k_kmeans_point_upd: incremental single-point centroid update
//    (ClusterIndex.add/remove): c = (c*cnt +/- x) / (cnt +/- 1).
//
// All points/centroid inputs bf16 (the corpus dtype); accumulation fp32.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x2k __attribute__((ext_vector_type(2)));

namespace kmeans_detail {

typedef __bf16 bf16x8k __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float dot_short8(short8v a, short8v b, float acc) {
  bf16x8k av = (bf16x8k)a, bv = (bf16x8k)b;  // same-size vector cast = bitcast
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    bf16x2k pa = {av[2 * j], av[2 * j + 1]};
    bf16x2k pb = {bv[2 * j], bv[2 * j + 1]};
    acc = __builtin_amdgcn_fdot2_f32_bf16(pa, pb, acc, false);
  }
  return acc;
}

}  // namespace kmeans_detail

// NV = d / 512 (dims per lane / 8). One wave per point.
template <int NV>
__global__ __launch_bounds__(256) void k_kmeans_assign(
    const unsigned short* __restrict__ x,     // [n][d] bf16
    const unsigned short* __restrict__ cent,  // [k][d] bf16
    const float* __restrict__ cnorm2,         // [k]
    long long n, int k, int d,
    int* __restrict__ assign, float* __restrict__ d2out) {
  using namespace kmeans_detail;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long long w0 = (long long)blockIdx.x * (blockDim.x / WAVE) + wid;
  const long long tw = (long long)gridDim.x * (blockDim.x / WAVE);

  for (long long p = w0; p < n; p += tw) {
    short8v xa[NV];
    float xsq = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      xa[v] = *reinterpret_cast<const short8v*>(
          x + p * d + (long long)(v * WAVE + lane) * 8);
      xsq = dot_short8(xa[v], xa[v], xsq);
    }
    xsq = wave_reduce_sum(xsq);

    float best = 3.4e38f;
    int besti = 0;
    for (int c = 0; c < k; ++c) {
      float dot = 0.f;
#pragma unroll
      for (int v = 0; v < NV; ++v) {
        short8v ca = *reinterpret_cast<const short8v*>(
            cent + (long long)c * d + (long long)(v * WAVE + lane) * 8);
        dot = dot_short8(xa[v], ca, dot);
      }
      dot = wave_reduce_sum(dot);
      float score = cnorm2[c] - 2.0f * dot;  // d2 - xsq
      if (score < best) {
        best = score;
        besti = c;
      }
    }
    if (lane == 0) {
      assign[p] = besti;
      float d2 = xsq + best;
      d2out[p] = d2 > 0.f ? d2 : 0.f;
    }
  }
}

__global__ __launch_bounds__(256) void k_kmeans_accum(
    const unsigned short* __restrict__ x, const int* __restrict__ assign,
    long long n, int d, float* __restrict__ sums, int* __restrict__ counts) {
  for (long long p = blockIdx.x; p < n; p += gridDim.x) {
    const int a = assign[p];
    float* dst = sums + (long long)a * d;
    const unsigned short* src = x + (long long)p * d;
    for (int i = threadIdx.x; i < d; i += blockDim.x) {
      atomicAdd(dst + i, bf16_bits_to_f32(src[i]));
    }
    if (threadIdx.x == 0) atomicAdd(counts + a, 1);
  }
}

// one wave per cluster: finalize + drift^2
__global__ __launch_bounds__(256) void k_kmeans_finalize(
    const float* __restrict__ sums, const int* __restrict__ counts,
    const float* __restrict__ old_c, float* __restrict__ new_c,
    float* __restrict__ drift2, int k, int d) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int c = blockIdx.x * (blockDim.x / WAVE) + wid;
  if (c >= k) return;
  const int cnt = counts[c];
  float dr = 0.f;
  for (int i = lane; i < d; i += WAVE) {
    float nv = cnt > 0 ? sums[(long long)c * d + i] / (float)cnt
                       : old_c[(long long)c * d + i];
    float ov = old_c[(long long)c * d + i];
    new_c[(long long)c * d + i] = nv;
    dr += (nv - ov) * (nv - ov);
  }
  dr = wave_reduce_sum(dr);
  if (lane == 0) drift2[c] = dr;
}

// k-means++ seeding step: d2 = min(d2, |x_p - c|^2), one new centroid.
template <int NV>
__global__ __launch_bounds__(256) void k_kmeanspp_update(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ c,
    float cn2, long long n, int d, float* __restrict__ d2) {
  using namespace kmeans_detail;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long long w0 = (long long)blockIdx.x * (blockDim.x / WAVE) + wid;
  const long long tw = (long long)gridDim.x * (blockDim.x / WAVE);
  for (long long p = w0; p < n; p += tw) {
    float dot = 0.f, xsq = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      short8v xa = *reinterpret_cast<const short8v*>(
          x + p * d + (long long)(v * WAVE + lane) * 8);
      short8v ca = *reinterpret_cast<const short8v*>(
          c + (long long)(v * WAVE + lane) * 8);
      dot = dot_short8(xa, ca, dot);
      xsq = dot_short8(xa, xa, xsq);
    }
    dot = wave_reduce_sum(dot);
    xsq = wave_reduce_sum(xsq);
    if (lane == 0) {
      float nd = xsq + cn2 - 2.0f * dot;
      nd = nd > 0.f ? nd : 0.f;
      if (nd < d2[p]) d2[p] = nd;
    }
  }
}

// incremental single-point update: sign=+1 add, -1 remove.
__global__ void k_kmeans_point_upd(float* __restrict__ cent,
                                   int* __restrict__ counts,
                                   const unsigned short* __restrict__ xv,
                                   int c, int d, int sign) {
  const int cnt_old = counts[c];
  const int cnt_new = cnt_old + sign;
  if (cnt_new <= 0) {
    if (threadIdx.x == 0 && blockIdx.x == 0) counts[c] = 0;
    return;
  }
  const float inv = 1.0f / (float)cnt_new;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < d;
       i += gridDim.x * blockDim.x) {
    float cur = cent[(long long)c * d + i];
    cent[(long long)c * d + i] =
        (cur * (float)cnt_old + (float)sign * bf16_bits_to_f32(xv[i])) * inv;
  }
  if (threadIdx.x == 0 && blockIdx.x == 0) counts[c] = cnt_new;
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------
static inline hipStream_t km_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define KM_CHECK_X(x)                                                        \
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&            \
                  x.scalar_type() == at::kBFloat16,                          \
              "kmeans: expected contiguous 2D bf16 CUDA tensor")

std::tuple<at::Tensor, at::Tensor> kmeans_assign(at::Tensor x, at::Tensor cent,
                                                 at::Tensor cnorm2) {
  KM_CHECK_X(x);
  KM_CHECK_X(cent);
  long long n = x.size(0);
  int d = (int)x.size(1), k = (int)cent.size(0);
  TORCH_CHECK(cent.size(1) == d, "kmeans_assign: dim mismatch");
  TORCH_CHECK(d % 512 == 0 && d <= 4096, "kmeans_assign: d % 512 == 0, <= 4096");
  TORCH_CHECK(cnorm2.scalar_type() == at::kFloat && cnorm2.numel() == k);
  auto assign = at::empty({n}, x.options().dtype(at::kInt));
  auto d2 = at::empty({n}, x.options().dtype(at::kFloat));
  int blocks = (int)std::min<long long>((n + 3) / 4, 8192);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(std::max(blocks, 1)), dim3(256), 0,
                       km_stream(), (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)cent.data_ptr(),
                       cnorm2.data_ptr<float>(), n, k, d,
                       assign.data_ptr<int>(), d2.data_ptr<float>());
  };
  switch (d / 512) {
    case 1: launch(k_kmeans_assign<1>); break;
    case 2: launch(k_kmeans_assign<2>); break;
    case 4: launch(k_kmeans_assign<4>); break;
    case 8: launch(k_kmeans_assign<8>); break;
    default: TORCH_CHECK(false, "kmeans_assign: unsupported d");
  }
  HIP_CHECK_LAST();
  return {assign, d2};
}

std::tuple<at::Tensor, at::Tensor> kmeans_accum(at::Tensor x, at::Tensor assign,
                                                long long k) {
  KM_CHECK_X(x);
  TORCH_CHECK(assign.scalar_type() == at::kInt && assign.numel() == x.size(0));
  long long n = x.size(0);
  int d = (int)x.size(1);
  auto sums = at::zeros({k, d}, x.options().dtype(at::kFloat));
  auto counts = at::zeros({k}, x.options().dtype(at::kInt));
  int blocks = (int)std::min<long long>(n, 4096);
  hipLaunchKernelGGL(k_kmeans_accum, dim3(std::max(blocks, 1)), dim3(256), 0,
                     km_stream(), (const unsigned short*)x.data_ptr(),
                     assign.data_ptr<int>(), n, d, sums.data_ptr<float>(),
                     counts.data_ptr<int>());
  HIP_CHECK_LAST();
  return {sums, counts};
}

std::tuple<at::Tensor, at::Tensor> kmeans_finalize(at::Tensor sums,
                                                   at::Tensor counts,
                                                   at::Tensor old_c) {
  int k = (int)sums.size(0), d = (int)sums.size(1);
  auto new_c = at::empty_like(old_c);
  auto drift2 = at::empty({k}, sums.options());
  int wpb = 4;
  int blocks = (k + wpb - 1) / wpb;
  hipLaunchKernelGGL(k_kmeans_finalize, dim3(std::max(blocks, 1)),
                     dim3(wpb * WAVE), 0, km_stream(),
                     sums.data_ptr<float>(), counts.data_ptr<int>(),
                     old_c.data_ptr<float>(), new_c.data_ptr<float>(),
                     drift2.data_ptr<float>(), k, d);
  HIP_CHECK_LAST();
  return {new_c, drift2};
}

void kmeanspp_update(at::Tensor x, at::Tensor c, double cn2, at::Tensor d2) {
  KM_CHECK_X(x);
  long long n = x.size(0);
  int d = (int)x.size(1);
  TORCH_CHECK(d % 512 == 0 && d <= 4096);
  int blocks = (int)std::min<long long>((n + 3) / 4, 8192);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(std::max(blocks, 1)), dim3(256), 0,
                       km_stream(), (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)c.data_ptr(), (float)cn2, n, d,
                       d2.data_ptr<float>());
  };
  switch (d / 512) {
    case 1: launch(k_kmeanspp_update<1>); break;
    case 2: launch(k_kmeanspp_update<2>); break;
    case 4: launch(k_kmeanspp_update<4>); break;
    case 8: launch(k_kmeanspp_update<8>); break;
    default: TORCH_CHECK(false, "kmeanspp_update: unsupported d");
  }
  HIP_CHECK_LAST();
}

void kmeans_point_update(at::Tensor cent, at::Tensor counts, at::Tensor xv,
                         long long c, long long sign) {
  int d = (int)cent.size(1);
  hipLaunchKernelGGL(k_kmeans_point_upd, dim3(4), dim3(256), 0, km_stream(),
                     cent.data_ptr<float>(), counts.data_ptr<int>(),
                     (const unsigned short*)xv.data_ptr(), (int)c, d,
                     (int)sign);
  HIP_CHECK_LAST();
}
