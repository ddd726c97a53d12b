// Fused single-token decoder step for Heimdall (Qwen2-shape, batch 1).
//
// The eager decode loop launches ~150 kernels per token and even the
// hipGraph replay is bound by kernel-granularity (~180 x ~25 us); this
// kernel runs ALL decoder layers for one token in a single cooperative
// launch (grid-wide sync between stages), reading each weight exactly
// once per token. lm_head + sampling stay in torch (hipBLASLt GEMV).
//
// Replaces the reference's llama.cpp decode runtime (pkg/localllm,
// pkg/heimdall scheduler.go) with an MI355X-native path.
//
// Math: bf16 weights, fp32 accumulation, RoPE rotate-half convention
// matching models/heimdall.py:_rope.
//
// Stage layout per layer (5 grid syncs):
//   S2 rmsnorm1(LDS, redundant per WG) + qkv GEMV pairs + RoPE + cache
//   S3 attention (wave per head, two-pass softmax via LDS scores)
//   S4 o-proj GEMV + residual
//   S5 rmsnorm2 + gate/up GEMV + SiLU*up -> h
//   S6 down GEMV + residual
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_cooperative_groups.h>
#include "common.h"

namespace cg = cooperative_groups;

typedef __bf16 bf16x2d __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x8d __attribute__((ext_vector_type(8)));

#define DWG 256          // threads per workgroup
#define DGRID 64         // workgroups (measured best: grid.sync cost grows with WGs, GEMV stages saturate by ~64)

// per-layer device pointers (filled host-side into an int64 tensor)
struct LayerPtrs {
  const unsigned short *ln1_w, *q_w, *q_b, *k_w, *k_b, *v_w, *v_b, *o_w;
  const unsigned short *ln2_w, *gate_w, *up_w, *down_w;
  unsigned short *cache_k, *cache_v;   // [nkv, max_len, hd]
};

struct DecodeArgs {
  const LayerPtrs* layers;
  int n_layers;
  int hidden;        // 896
  int n_heads;       // 14
  int n_kv;          // 2
  int hd;            // 64
  int inter;         // 4864
  int max_len;
  float rms_eps;
  const float* rope_cos;   // [max_pos, hd]
  const float* rope_sin;
  float* x;          // [hidden] fp32 hidden state (in/out)
  float* q;          // [hidden] scratch
  float* attn;       // [hidden] scratch
  float* h;          // [inter] scratch
  int pos;           // current position (token index)
};

// dot(bf16 row, fp32 vec in LDS) over `n` (multiple of 8), one WAVE,
// fp32 accumulate. Lane reads 16 B per step (b128, coalesced: the wave
// streams 1 KB/iteration) — 4 B/lane loads measured only ~140 GB/s
// aggregate (latency-bound short loops).
__device__ __forceinline__ float wave_dot_bf16(
    const unsigned short* __restrict__ w, const float* __restrict__ r,
    int n, int lane) {
  float acc = 0.f;
#pragma unroll 2
  for (int k = lane * 8; k < n; k += WAVE * 8) {
    bf16x8d wv = *reinterpret_cast<const bf16x8d*>(w + k);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += (float)wv[j] * r[k + j];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, WAVE);
  return acc;   // every lane holds the full sum
}

// four dots in flight (two row pairs): per k-iteration the lane issues
// 4 independent b128 loads, so the wave is not latency-serialized on a
// single row chain (single-dot loops measured ~220 GB/s aggregate).
__device__ __forceinline__ void wave_dot2_bf16(
    const unsigned short* __restrict__ w0, const unsigned short* __restrict__ w1,
    const float* __restrict__ r, int n, int lane, float out[2]) {
  float a0 = 0.f, a1 = 0.f;
#pragma unroll 2
  for (int k = lane * 8; k < n; k += WAVE * 8) {
    bf16x8d v0 = *reinterpret_cast<const bf16x8d*>(w0 + k);
    bf16x8d v1 = *reinterpret_cast<const bf16x8d*>(w1 + k);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float rv = r[k + j];
      a0 += (float)v0[j] * rv;
      a1 += (float)v1[j] * rv;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    a0 += __shfl_xor(a0, off, WAVE);
    a1 += __shfl_xor(a1, off, WAVE);
  }
  out[0] = a0; out[1] = a1;
}

__device__ __forceinline__ void wave_dot4_bf16(
    const unsigned short* __restrict__ w0, const unsigned short* __restrict__ w1,
    const unsigned short* __restrict__ w2, const unsigned short* __restrict__ w3,
    const float* __restrict__ r, int n, int lane, float out[4]) {
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll 2
  for (int k = lane * 8; k < n; k += WAVE * 8) {
    bf16x8d v0 = *reinterpret_cast<const bf16x8d*>(w0 + k);
    bf16x8d v1 = *reinterpret_cast<const bf16x8d*>(w1 + k);
    bf16x8d v2 = *reinterpret_cast<const bf16x8d*>(w2 + k);
    bf16x8d v3 = *reinterpret_cast<const bf16x8d*>(w3 + k);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float rv = r[k + j];
      a0 += (float)v0[j] * rv;
      a1 += (float)v1[j] * rv;
      a2 += (float)v2[j] * rv;
      a3 += (float)v3[j] * rv;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    a0 += __shfl_xor(a0, off, WAVE);
    a1 += __shfl_xor(a1, off, WAVE);
    a2 += __shfl_xor(a2, off, WAVE);
    a3 += __shfl_xor(a3, off, WAVE);
  }
  out[0] = a0; out[1] = a1; out[2] = a2; out[3] = a3;
}

// rmsnorm of x[hidden] scaled by w -> LDS r[hidden]; redundant per WG.
__device__ __forceinline__ void wg_rmsnorm(
    const float* __restrict__ x, const unsigned short* __restrict__ w,
    float* __restrict__ r_lds, int hidden, float eps, int tid) {
  __shared__ float s_ssq[DWG / WAVE];
  float ssq = 0.f;
  for (int i = tid; i < hidden; i += DWG) ssq += x[i] * x[i];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    ssq += __shfl_xor(ssq, off, WAVE);
  if ((tid & (WAVE - 1)) == 0) s_ssq[tid / WAVE] = ssq;
  __syncthreads();
  float tot = 0.f;
#pragma unroll
  for (int i = 0; i < DWG / WAVE; ++i) tot += s_ssq[i];
  float inv = rsqrtf(tot / hidden + eps);
  for (int i = tid; i < hidden; i += DWG)
    r_lds[i] = x[i] * inv * bf16_bits_to_f32(w[i]);
  __syncthreads();
}

__global__ void __launch_bounds__(DWG, 1) k_decode_step(DecodeArgs a) {
  cg::grid_group grid = cg::this_grid();
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;                      // wave in WG (0..3)
  const int gwave = blockIdx.x * (DWG / WAVE) + wid;  // global wave id
  const int n_gwaves = gridDim.x * (DWG / WAVE);

  extern __shared__ float smem[];                  // r[hidden] | h[inter]
  float* r_lds = smem;

  const int H = a.hidden, HD = a.hd, NH = a.n_heads, NKV = a.n_kv;
  const int q_rows = NH * HD;                      // 896
  const int kv_rows = NKV * HD;                    // 128
  const float* cosp = a.rope_cos + (long long)a.pos * HD;
  const float* sinp = a.rope_sin + (long long)a.pos * HD;

  for (int li = 0; li < a.n_layers; ++li) {
    const LayerPtrs L = a.layers[li];

    // ---- S2: rmsnorm1 + fused qkv GEMV (RoPE'd) + cache append ----
    wg_rmsnorm(a.x, L.ln1_w, r_lds, H, a.rms_eps, tid);
    // pair units: q pairs [0, NH*HD/2) -> rows (h*HD+j, h*HD+j+HD/2);
    // k pairs [NH*HD/2, NH*HD/2 + NKV*HD/2); v rows as HD/2-sized pairs
    const int q_pairs = q_rows / 2, k_pairs = kv_rows / 2,
              v_pairs = kv_rows / 2;
#pragma unroll 2
    for (int p = gwave; p < q_pairs + k_pairs + v_pairs; p += n_gwaves) {
      if (p < q_pairs) {
        int head = p / (HD / 2), j = p % (HD / 2);
        int r1 = head * HD + j, r2 = r1 + HD / 2;
        float d1 = wave_dot_bf16(L.q_w + (long long)r1 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.q_b[r1]);
        float d2 = wave_dot_bf16(L.q_w + (long long)r2 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.q_b[r2]);
        if (lane == 0) {
          a.q[r1] = d1 * cosp[j] - d2 * sinp[j];
          a.q[r2] = d2 * cosp[j + HD / 2] + d1 * sinp[j + HD / 2];
        }
      } else if (p < q_pairs + k_pairs) {
        int pk = p - q_pairs;
        int kvh = pk / (HD / 2), j = pk % (HD / 2);
        int r1 = kvh * HD + j, r2 = r1 + HD / 2;
        float d1 = wave_dot_bf16(L.k_w + (long long)r1 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.k_b[r1]);
        float d2 = wave_dot_bf16(L.k_w + (long long)r2 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.k_b[r2]);
        if (lane == 0) {
          long long base = ((long long)kvh * a.max_len + a.pos) * HD;
          L.cache_k[base + j] =
              f32_to_bf16_bits(d1 * cosp[j] - d2 * sinp[j]);
          L.cache_k[base + j + HD / 2] =
              f32_to_bf16_bits(d2 * cosp[j + HD / 2] + d1 * sinp[j + HD / 2]);
        }
      } else {
        int pv = p - q_pairs - k_pairs;
        int kvh = pv / (HD / 2), j = pv % (HD / 2);
        int r1 = kvh * HD + j, r2 = r1 + HD / 2;
        float d1 = wave_dot_bf16(L.v_w + (long long)r1 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.v_b[r1]);
        float d2 = wave_dot_bf16(L.v_w + (long long)r2 * H, r_lds, H, lane)
                   + bf16_bits_to_f32(L.v_b[r2]);
        if (lane == 0) {
          long long base = ((long long)kvh * a.max_len + a.pos) * HD;
          L.cache_v[base + j] = f32_to_bf16_bits(d1);
          L.cache_v[base + j + HD / 2] = f32_to_bf16_bits(d2);
        }
      }
    }
    grid.sync();

    // ---- S3: attention, one wave per head ----
    // two-pass: scores into LDS (positions strided by lane), softmax,
    // then out[d] per lane. LDS reuse: scores live in r_lds.
    if (gwave < NH) {
      const int head = gwave;
      const int kvh = head / (NH / NKV);
      const unsigned short* K = L.cache_k + (long long)kvh * a.max_len * HD;
      const unsigned short* V = L.cache_v + (long long)kvh * a.max_len * HD;
      const int T = a.pos + 1;
      const float* qv = a.q + head * HD;
      const float scale = rsqrtf((float)HD);
      float mx = -1e30f;
      // scores: lane handles t = lane, lane+64, ...
      float* sc = r_lds + wid * 2048;   // per-wave region (<=2048 pos)
      for (int t = lane; t < T; t += WAVE) {
        const unsigned short* kr = K + (long long)t * HD;
        float s = 0.f;
#pragma unroll
        for (int d = 0; d < 64; d += 2) {
          bf16x2d kv2 = *reinterpret_cast<const bf16x2d*>(kr + d);
          s += (float)kv2[0] * qv[d] + (float)kv2[1] * qv[d + 1];
        }
        s *= scale;
        sc[t] = s;
        mx = fmaxf(mx, s);
      }
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      float se = 0.f;
      for (int t = lane; t < T; t += WAVE) {
        float e = __expf(sc[t] - mx);
        sc[t] = e;
        se += e;
      }
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        se += __shfl_xor(se, off, WAVE);
      const float inv = 1.f / se;
      // out[d]: lane d accumulates over t (V column read, stride HD)
      if (lane < HD) {
        float o = 0.f;
        for (int t = 0; t < T; ++t)
          o += sc[t] * (float)(*reinterpret_cast<const __bf16*>(
              V + (long long)t * HD + lane));
        a.attn[head * HD + lane] = o * inv;
      }
    }
    grid.sync();

    // ---- S4: o-proj + residual (wave per row) ----
    // stage attn into LDS (fp32) once per WG
    for (int i = tid; i < H; i += DWG) r_lds[i] = a.attn[i];
    __syncthreads();
    for (int row = gwave; row < H; row += n_gwaves * 2) {
      int row2 = row + n_gwaves;
      if (row2 < H) {
        float d2[2];
        wave_dot2_bf16(L.o_w + (long long)row * H,
                       L.o_w + (long long)row2 * H, r_lds, H, lane, d2);
        if (lane == 0) { a.x[row] += d2[0]; a.x[row2] += d2[1]; }
      } else {
        float d = wave_dot_bf16(L.o_w + (long long)row * H, r_lds, H, lane);
        if (lane == 0) a.x[row] += d;
      }
    }
    grid.sync();

    // ---- S5: rmsnorm2 + gate/up + SiLU (2 rows x 2 mats in flight) ----
    wg_rmsnorm(a.x, L.ln2_w, r_lds, H, a.rms_eps, tid);
#pragma unroll 2
    for (int row = gwave; row < a.inter; row += n_gwaves * 2) {
      int row2 = row + n_gwaves;
      float d[4];
      if (row2 < a.inter) {
        wave_dot4_bf16(L.gate_w + (long long)row * H,
                       L.up_w + (long long)row * H,
                       L.gate_w + (long long)row2 * H,
                       L.up_w + (long long)row2 * H, r_lds, H, lane, d);
        if (lane == 0) {
          a.h[row] = (d[0] / (1.f + __expf(-d[0]))) * d[1];
          a.h[row2] = (d[2] / (1.f + __expf(-d[2]))) * d[3];
        }
      } else {
        float g = wave_dot_bf16(L.gate_w + (long long)row * H, r_lds, H, lane);
        float u = wave_dot_bf16(L.up_w + (long long)row * H, r_lds, H, lane);
        if (lane == 0) a.h[row] = (g / (1.f + __expf(-g))) * u;
      }
    }
    grid.sync();

    // ---- S6: down-proj + residual ----
    // h is 4864 fp32 = 19.5 KB -> LDS per WG
    for (int i = tid; i < a.inter; i += DWG) r_lds[i] = a.h[i];
    __syncthreads();
    for (int row = gwave; row < H; row += n_gwaves * 2) {
      int row2 = row + n_gwaves;
      if (row2 < H) {
        float d[2];
        wave_dot2_bf16(L.down_w + (long long)row * a.inter,
                       L.down_w + (long long)row2 * a.inter,
                       r_lds, a.inter, lane, d);
        if (lane == 0) {
          a.x[row] += d[0];
          a.x[row2] += d[1];
        }
      } else {
        float d = wave_dot_bf16(L.down_w + (long long)row * a.inter,
                                r_lds, a.inter, lane);
        if (lane == 0) a.x[row] += d;
      }
    }
    grid.sync();
  }
}

// ---------------------------------------------------------------------------
// host wrapper
// ---------------------------------------------------------------------------
void decode_step(at::Tensor layer_ptrs,  // [n_layers, 14] int64 (LayerPtrs)
                 at::Tensor x, at::Tensor q, at::Tensor attn, at::Tensor h,
                 at::Tensor rope_cos, at::Tensor rope_sin,
                 long long n_layers, long long hidden, long long n_heads,
                 long long n_kv, long long hd, long long inter,
                 long long max_len, double rms_eps, long long pos) {
  TORCH_CHECK(layer_ptrs.is_cuda() && layer_ptrs.scalar_type() == at::kLong
                  && layer_ptrs.is_contiguous(),
              "layer_ptrs must be contiguous CUDA int64");
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.is_contiguous());
  TORCH_CHECK(rope_cos.scalar_type() == at::kFloat &&
              rope_cos.is_contiguous());
  TORCH_CHECK(max_len <= 2048, "fused decode supports max_len <= 2048");
  TORCH_CHECK(hd == 64 && hidden % 2 == 0);

  DecodeArgs a;
  a.layers = reinterpret_cast<const LayerPtrs*>(layer_ptrs.data_ptr());
  a.n_layers = (int)n_layers;
  a.hidden = (int)hidden;
  a.n_heads = (int)n_heads;
  a.n_kv = (int)n_kv;
  a.hd = (int)hd;
  a.inter = (int)inter;
  a.max_len = (int)max_len;
  a.rms_eps = (float)rms_eps;
  a.rope_cos = rope_cos.data_ptr<float>();
  a.rope_sin = rope_sin.data_ptr<float>();
  a.x = x.data_ptr<float>();
  a.q = q.data_ptr<float>();
  a.attn = attn.data_ptr<float>();
  a.h = h.data_ptr<float>();
  a.pos = (int)pos;

  // LDS: max(hidden, inter, 4 waves * 2048 scores) floats
  size_t lds = sizeof(float) *
      std::max<long long>(std::max<long long>(hidden, inter),
                          4LL * 2048);
  auto stream = at::hip::getCurrentHIPStream().stream();
  // Grid size tunes the grid.sync cost (dominant at 256 WGs: ~35 us per
  // sync); overridable for experiments via NORNICDB_DECODE_GRID.
  int grid = DGRID;
  if (const char* g = getenv("NORNICDB_DECODE_GRID")) grid = atoi(g);
  void* args[] = {&a};
  hipError_t err = hipLaunchCooperativeKernel(
      reinterpret_cast<void*>(&k_decode_step), dim3(grid), dim3(DWG),
      args, lds, stream);
  TORCH_CHECK(err == hipSuccess, "decode_step launch failed: ",
              hipGetErrorString(err));
  HIP_CHECK_LAST();
}
