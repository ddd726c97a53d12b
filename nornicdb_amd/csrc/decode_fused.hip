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

#define DWG 512          // threads per workgroup (8 waves = 2/SIMD: the
                         // GEMV stages are latency-bound at 1 wave/SIMD —
                         // measured 2.17 ms/token at DWG 256, grid 64)
#define DGRID 64         // workgroups (grid.sync cost grows with WGs; GEMV
                         // stages gain more from waves/CU than from CUs)

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

// gate+up dots for FOUR rows at once (8 independent b128 streams per
// k-iteration — with H=896 the k-loop is only 2 rounds, so in-flight
// loads per wave, not loop depth, set the memory-level parallelism).
// Per-row accumulation order matches wave_dot4 exactly (bitwise).
__device__ __forceinline__ void wave_dot_gu4_bf16(
    const unsigned short* __restrict__ gw,
    const unsigned short* __restrict__ uw, const int rows[4],
    const float* __restrict__ r, int n, int lane, float g[4], float u[4]) {
  float ag[4] = {0.f, 0.f, 0.f, 0.f}, au[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll 1
  for (int k = lane * 8; k < n; k += WAVE * 8) {
    bf16x8d vg[4], vu[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      vg[i] = *reinterpret_cast<const bf16x8d*>(gw + (long long)rows[i] * n + k);
      vu[i] = *reinterpret_cast<const bf16x8d*>(uw + (long long)rows[i] * n + k);
    }
    float rv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) rv[j] = r[k + j];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ag[i] += (float)vg[i][j] * rv[j];
        au[i] += (float)vu[i][j] * rv[j];
      }
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      ag[i] += __shfl_xor(ag[i], off, WAVE);
      au[i] += __shfl_xor(au[i], off, WAVE);
    }
    g[i] = ag[i];
    u[i] = au[i];
  }
}

// two-row dot with a 4-deep staged k-loop (8 loads in flight): for long
// rows (down-proj K=4864) the plain dot2 keeps only 4 loads in flight.
// Accumulation order per row is identical to wave_dot2 (k ascending).
__device__ __forceinline__ void wave_dot2_k4_bf16(
    const unsigned short* __restrict__ w0,
    const unsigned short* __restrict__ w1, const float* __restrict__ r,
    int n, int lane, float out[2]) {
  float a0 = 0.f, a1 = 0.f;
  int k = lane * 8;
  const int step = WAVE * 8;
  for (; k + 3 * step < n; k += 4 * step) {
    bf16x8d v0[4], v1[4];
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      v0[s] = *reinterpret_cast<const bf16x8d*>(w0 + k + s * step);
      v1[s] = *reinterpret_cast<const bf16x8d*>(w1 + k + s * step);
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float rv = r[k + s * step + j];
        a0 += (float)v0[s][j] * rv;
        a1 += (float)v1[s][j] * rv;
      }
    }
  }
  for (; k < n; k += step) {
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
  out[0] = a0;
  out[1] = a1;
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

// All decoder layers for one token at position a.pos (5 grid-wide syncs
// per layer). Shared by the single-step kernel (host-side lm_head +
// sampling) and the multi-token greedy kernel below.
__device__ void run_layers(const DecodeArgs& a, cg::grid_group& grid,
                           float* r_lds, int tid, int lane, int wid,
                           int gwave, int n_gwaves) {
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
      // static LDS: 8 waves x 4096 positions = 128 KB on top of the
      // ~20 KB dynamic r_lds — gfx950 allows up to 160 KB per WG for
      // STATIC allocations (128 KB measured fine in the GEMM kernels),
      // while >=64 KB DYNAMIC makes the cooperative launch fail with
      // invalid argument (measured). 4096 = the Qwen2 max_position.
      static __shared__ float s_scores[(DWG / WAVE) * 4096];
      float* sc = s_scores + wid * 4096;   // per-wave region (<=4096 pos)
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

    // ---- S5: rmsnorm2 + gate/up + SiLU (4 rows x 2 mats in flight) ----
    wg_rmsnorm(a.x, L.ln2_w, r_lds, H, a.rms_eps, tid);
    for (int r0 = gwave; r0 < a.inter; r0 += n_gwaves * 4) {
      int rows[4];
      bool valid[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int rr = r0 + i * n_gwaves;
        valid[i] = rr < a.inter;
        rows[i] = valid[i] ? rr : 0;   // clamped rows read row 0, discarded
      }
      float g[4], u[4];
      wave_dot_gu4_bf16(L.gate_w, L.up_w, rows, r_lds, H, lane, g, u);
      if (lane == 0)
#pragma unroll
        for (int i = 0; i < 4; ++i)
          if (valid[i])
            a.h[rows[i]] = (g[i] / (1.f + __expf(-g[i]))) * u[i];
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
        wave_dot2_k4_bf16(L.down_w + (long long)row * a.inter,
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

__global__ void __launch_bounds__(DWG, 1) k_decode_step(DecodeArgs a) {
  cg::grid_group grid = cg::this_grid();
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;                      // wave in WG (0..3)
  const int gwave = blockIdx.x * (DWG / WAVE) + wid;  // global wave id
  const int n_gwaves = gridDim.x * (DWG / WAVE);
  extern __shared__ float smem[];                  // r[hidden] | h[inter]
  run_layers(a, grid, smem, tid, lane, wid, gwave, n_gwaves);
}

// ---------------------------------------------------------------------------
// multi-token greedy decode: the whole generation loop in ONE launch
// ---------------------------------------------------------------------------

// lm_head GEMV + running argmax over `vocab` rows. 8 rows in flight per
// wave: a single wave-per-row dot is latency-serialized (~14 dependent
// b128 rounds per 896-wide row); staging 8 row streams keeps ~2 MB of
// loads in flight across the grid and turns the 272 MB lm_head read
// bandwidth-bound. Every lane exits with the wave's (best, idx) —
// wave_dot reductions leave identical sums on all lanes.
__device__ void wave_lm_argmax(const unsigned short* __restrict__ w,
                               const float* __restrict__ r, int vocab,
                               int n, int lane, int gwave, int n_gwaves,
                               float& best, int& bi) {
  best = -3.4e38f;
  bi = 0x7fffffff;
  for (long long row0 = (long long)gwave * 8; row0 < vocab;
       row0 += (long long)n_gwaves * 8) {
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    const int nr = (int)((vocab - row0) < 8 ? (vocab - row0) : 8);
    if (nr == 8) {
      const unsigned short* w0 = w + row0 * n;
#pragma unroll 1
      for (int k = lane * 8; k < n; k += WAVE * 8) {
        bf16x8d v[8];
#pragma unroll
        for (int rr = 0; rr < 8; ++rr)
          v[rr] = *reinterpret_cast<const bf16x8d*>(w0 + (long long)rr * n + k);
        float rv[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) rv[j] = r[k + j];
#pragma unroll
        for (int rr = 0; rr < 8; ++rr)
#pragma unroll
          for (int j = 0; j < 8; ++j) acc[rr] += (float)v[rr][j] * rv[j];
      }
    } else {
      for (int rr = 0; rr < nr; ++rr) {
        const unsigned short* wr = w + (row0 + rr) * n;
        for (int k = lane * 8; k < n; k += WAVE * 8) {
          bf16x8d v = *reinterpret_cast<const bf16x8d*>(wr + k);
#pragma unroll
          for (int j = 0; j < 8; ++j) acc[rr] += (float)v[j] * r[k + j];
        }
      }
    }
#pragma unroll
    for (int rr = 0; rr < 8; ++rr)
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        acc[rr] += __shfl_xor(acc[rr], off, WAVE);
    for (int rr = 0; rr < nr; ++rr) {
      int row = (int)(row0 + rr);
      if (acc[rr] > best || (acc[rr] == best && row < bi)) {
        best = acc[rr];
        bi = row;
      }
    }
  }
}

struct DecodeTokArgs {
  DecodeArgs base;               // base.pos = position of start_tok
  const unsigned short* embed;   // [vocab, hidden] token embedding (bf16)
  const unsigned short* norm_w;  // final rmsnorm weight
  const unsigned short* lm_w;    // [vocab, hidden] (tied = embed for Qwen2)
  int vocab;
  int n_toks;                    // max tokens to emit this launch
  int start_tok;                 // token to feed at base.pos
  int eos;                       // -1: none; emitted then stops
  int* out;                      // [n_toks] emitted token ids
  int* n_done;                   // [1] number emitted (host zero-inits)
  float* pmax;                   // [gridDim] per-WG argmax partials
  int* pidx;                     // [gridDim]
};

// Greedy decode of up to n_toks tokens in ONE cooperative launch: the
// per-token host round-trip (launch + lm_head GEMV + argmax + .item()
// sync + embed) disappears; each token is embed -> run_layers ->
// final rmsnorm -> lm_head argmax (grid-reduced) -> feed back. EOS
// stops in-kernel (every block computes the identical winner, so the
// break is grid-uniform). 122 grid syncs/token at the default grid.
__global__ void __launch_bounds__(DWG, 1) k_decode_tokens(DecodeTokArgs t) {
  cg::grid_group grid = cg::this_grid();
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int gwave = blockIdx.x * (DWG / WAVE) + wid;
  const int n_gwaves = gridDim.x * (DWG / WAVE);
  extern __shared__ float smem[];
  __shared__ float sb[DWG / WAVE];
  __shared__ int si[DWG / WAVE + 1];

  DecodeArgs a = t.base;
  const int pos0 = a.pos;
  int tok = t.start_tok;
  for (int step = 0; step < t.n_toks; ++step) {
    a.pos = pos0 + step;
    if (a.pos >= a.max_len) break;
    // embed the current token (block 0 writes x; every a.x read below
    // sits between this sync and the partials sync, so no other block
    // can observe a half-written x)
    if (blockIdx.x == 0) {
      const unsigned short* e = t.embed + (long long)tok * a.hidden;
      for (int i = tid; i < a.hidden; i += DWG)
        a.x[i] = bf16_bits_to_f32(e[i]);
    }
    grid.sync();
    run_layers(a, grid, smem, tid, lane, wid, gwave, n_gwaves);
    wg_rmsnorm(a.x, t.norm_w, smem, a.hidden, a.rms_eps, tid);
    // round normed x to bf16 so the argmax bit-matches the host lm_head
    // path (step_logits casts to bf16 before its GEMV)
    for (int i = tid; i < a.hidden; i += DWG)
      smem[i] = bf16_bits_to_f32(f32_to_bf16_bits(smem[i]));
    __syncthreads();
    float best;
    int bi;
    wave_lm_argmax(t.lm_w, smem, t.vocab, a.hidden, lane, gwave,
                   n_gwaves, best, bi);
    if (lane == 0) { sb[wid] = best; si[wid] = bi; }
    __syncthreads();
    if (tid == 0) {
      float b = sb[0];
      int i = si[0];
      for (int w = 1; w < DWG / WAVE; ++w)
        if (sb[w] > b || (sb[w] == b && si[w] < i)) { b = sb[w]; i = si[w]; }
      t.pmax[blockIdx.x] = b;
      t.pidx[blockIdx.x] = i;
    }
    grid.sync();
    // grid-wide winner, reduced redundantly per block (gridDim reads)
    if (tid == 0) {
      float b = t.pmax[0];
      int i = t.pidx[0];
      for (int g = 1; g < (int)gridDim.x; ++g) {
        float pb = t.pmax[g];
        int pi = t.pidx[g];
        if (pb > b || (pb == b && pi < i)) { b = pb; i = pi; }
      }
      si[DWG / WAVE] = i;
    }
    __syncthreads();
    tok = si[DWG / WAVE];
    if (blockIdx.x == 0 && tid == 0) {
      t.out[step] = tok;
      *t.n_done = step + 1;
    }
    if (t.eos >= 0 && tok == t.eos) break;
  }
}

// ---------------------------------------------------------------------------
// grid-barrier microbenchmark: cg::grid.sync vs hand-rolled two-level
// barrier (per-XCD-group arrival counter + global generation), to see
// whether the cooperative sync cost at large grids is arrival contention
// (fixable) or fence cost (not).
// ---------------------------------------------------------------------------

struct BarrierLine {
  unsigned v;
  unsigned pad[31];    // one counter per 128 B cacheline
};

struct BarrierState {
  BarrierLine cnt[8];  // per-group arrival counters
  BarrierLine gcnt;    // group-leader arrival counter
  BarrierLine gen;     // generation (release flag)
};

__device__ __forceinline__ void grid_barrier2(BarrierState* bs, int bid,
                                              int nblocks, int tid,
                                              unsigned gen) {
  __threadfence();                    // release (agent scope)
  __syncthreads();
  if (tid == 0) {
    // monotonic counters (no resets, so a fast re-arrival can never race
    // a leader's reset): generation k's last group member sees arrival
    // k*gsz-1; the last group leader sees k*ngroups-1 and publishes k.
    const int ngroups = (nblocks + 31) / 32;
    const int g = bid >> 5;
    const int gsz = min(32, nblocks - (g << 5));
    if (__hip_atomic_fetch_add(&bs->cnt[g].v, 1u, __ATOMIC_ACQ_REL,
                               __HIP_MEMORY_SCOPE_AGENT) ==
        gen * (unsigned)gsz - 1u) {
      if (__hip_atomic_fetch_add(&bs->gcnt.v, 1u, __ATOMIC_ACQ_REL,
                                 __HIP_MEMORY_SCOPE_AGENT) ==
          gen * (unsigned)ngroups - 1u) {
        __hip_atomic_store(&bs->gen.v, gen, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
      }
    }
    while (__hip_atomic_load(&bs->gen.v, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_AGENT) < gen)
      __builtin_amdgcn_s_sleep(8);
  }
  __syncthreads();
  __threadfence();                    // acquire
}

__global__ void __launch_bounds__(256, 1) k_sync_bench(int iters, int which,
                                                       BarrierState* bs,
                                                       float* sink) {
  cg::grid_group grid = cg::this_grid();
  float acc = 0.f;
  if (which == 0) {
    for (int i = 0; i < iters; ++i) {
      acc += 1.f;
      grid.sync();
    }
  } else {
    for (int i = 0; i < iters; ++i) {
      acc += 1.f;
      grid_barrier2(bs, blockIdx.x, gridDim.x, threadIdx.x, (unsigned)(i + 1));
    }
  }
  if (acc < 0.f) *sink = acc;
}

double sync_bench(long long iters, long long which, long long grid,
                  at::Tensor scratch) {
  TORCH_CHECK(scratch.is_cuda() && scratch.numel() >= 384);
  scratch.zero_();
  auto stream = at::hip::getCurrentHIPStream().stream();
  int it = (int)iters, wh = (int)which;
  BarrierState* bs = reinterpret_cast<BarrierState*>(scratch.data_ptr());
  float* sink = reinterpret_cast<float*>(scratch.data_ptr()) + 352;
  void* args[] = {&it, &wh, &bs, &sink};
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  (void)hipEventRecord(e0, stream);
  hipError_t err = hipLaunchCooperativeKernel(
      reinterpret_cast<void*>(&k_sync_bench), dim3((uint32_t)grid), dim3(256),
      args, 0, stream);
  TORCH_CHECK(err == hipSuccess, "sync_bench launch failed: ",
              hipGetErrorString(err));
  (void)hipEventRecord(e1, stream);
  (void)hipEventSynchronize(e1);
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, e0, e1);
  (void)hipEventDestroy(e0);
  (void)hipEventDestroy(e1);
  return (double)ms;
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
  TORCH_CHECK(max_len <= 4096, "fused decode supports max_len <= 4096");
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

  // LDS (dynamic): max(hidden, inter) floats; scores are static
  size_t lds = sizeof(float) *
      std::max<long long>(hidden, inter);
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

void decode_tokens(at::Tensor layer_ptrs, at::Tensor x, at::Tensor q,
                   at::Tensor attn, at::Tensor h, at::Tensor rope_cos,
                   at::Tensor rope_sin, at::Tensor embed_w,
                   at::Tensor norm_w, at::Tensor lm_w, at::Tensor out,
                   at::Tensor n_done, at::Tensor pmax, at::Tensor pidx,
                   long long n_layers, long long hidden, long long n_heads,
                   long long n_kv, long long hd, long long inter,
                   long long max_len, double rms_eps, long long pos,
                   long long start_tok, long long n_toks, long long eos) {
  TORCH_CHECK(layer_ptrs.is_cuda() && layer_ptrs.scalar_type() == at::kLong
                  && layer_ptrs.is_contiguous(),
              "layer_ptrs must be contiguous CUDA int64");
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.is_contiguous());
  TORCH_CHECK(embed_w.scalar_type() == at::kBFloat16 &&
              embed_w.is_contiguous() && embed_w.dim() == 2 &&
              embed_w.size(1) == hidden);
  TORCH_CHECK(lm_w.scalar_type() == at::kBFloat16 && lm_w.is_contiguous()
                  && lm_w.sizes() == embed_w.sizes(),
              "lm_w must match embed_w shape");
  TORCH_CHECK(norm_w.scalar_type() == at::kBFloat16 &&
              norm_w.is_contiguous() && norm_w.numel() == hidden);
  TORCH_CHECK(out.scalar_type() == at::kInt && out.is_contiguous() &&
              out.numel() >= n_toks);
  TORCH_CHECK(n_done.scalar_type() == at::kInt && n_done.numel() >= 1);
  TORCH_CHECK(max_len <= 4096, "fused decode supports max_len <= 4096");
  TORCH_CHECK(hd == 64 && hidden % 2 == 0);

  DecodeTokArgs t;
  t.base.layers = reinterpret_cast<const LayerPtrs*>(layer_ptrs.data_ptr());
  t.base.n_layers = (int)n_layers;
  t.base.hidden = (int)hidden;
  t.base.n_heads = (int)n_heads;
  t.base.n_kv = (int)n_kv;
  t.base.hd = (int)hd;
  t.base.inter = (int)inter;
  t.base.max_len = (int)max_len;
  t.base.rms_eps = (float)rms_eps;
  t.base.rope_cos = rope_cos.data_ptr<float>();
  t.base.rope_sin = rope_sin.data_ptr<float>();
  t.base.x = x.data_ptr<float>();
  t.base.q = q.data_ptr<float>();
  t.base.attn = attn.data_ptr<float>();
  t.base.h = h.data_ptr<float>();
  t.base.pos = (int)pos;
  t.embed = reinterpret_cast<const unsigned short*>(embed_w.data_ptr());
  t.norm_w = reinterpret_cast<const unsigned short*>(norm_w.data_ptr());
  t.lm_w = reinterpret_cast<const unsigned short*>(lm_w.data_ptr());
  t.vocab = (int)embed_w.size(0);
  t.n_toks = (int)n_toks;
  t.start_tok = (int)start_tok;
  t.eos = (int)eos;
  t.out = out.data_ptr<int>();
  t.n_done = n_done.data_ptr<int>();
  t.pmax = pmax.data_ptr<float>();
  t.pidx = pidx.data_ptr<int>();
  TORCH_CHECK(start_tok >= 0 && start_tok < t.vocab);

  size_t lds = sizeof(float) *
      std::max<long long>(hidden, inter);
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = DGRID;
  if (const char* g = getenv("NORNICDB_DECODE_GRID")) grid = atoi(g);
  TORCH_CHECK(pmax.numel() >= grid && pidx.numel() >= grid,
              "pmax/pidx scratch smaller than grid");
  void* args[] = {&t};
  hipError_t err = hipLaunchCooperativeKernel(
      reinterpret_cast<void*>(&k_decode_tokens), dim3(grid), dim3(DWG),
      args, lds, stream);
  TORCH_CHECK(err == hipSuccess, "decode_tokens launch failed: ",
              hipGetErrorString(err));
  HIP_CHECK_LAST();
}
