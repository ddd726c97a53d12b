// Standalone ablation probe for the fused kNN MFMA kernel (no torch deps).
// Compile: hipcc --offload-arch=gfx950 -O3 -std=c++17 knn_ablate.hip -o knn_ablate
// Variants (template<int V>):
//   0 = full kernel (stage A+B, MFMA, epilogue scan)
//   1 = no epilogue scan (C still written to LDS, kept live)
//   2 = no B staging (stale B tile)
//   3 = no A staging (stale A tile)
//   4 = MFMA only (no staging, no epilogue)
//   5 = full + threshold-gated epilogue (skip column scan when the LDS
//       chunk's max can't beat this thread's current kth best)
//   6 = full + TRANSPOSED epilogue chunk [col][row] -> float4 writes and
//       contiguous float4 scans per thread
//   7 = V6 + double-buffered LDS 2-phase loop (stage next tile before MFMA,
//       counted barrier per K-step)
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

#define WAVE 64
typedef short short8v __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
#define G_AS __attribute__((address_space(1)))
#define L_AS __attribute__((address_space(3)))

#define BM 64
#define BN 256
#define BK 64
#define NT 256
#define KC 12
#define SCH 32
#define SSTR (BN + 4)
#define TSTR 36   // transposed chunk row stride (32 + 4 pad)

__global__ void fill_rand(unsigned short* x, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned long long h = i * 0x9e3779b97f4a7c15ULL;
    h ^= h >> 33; h *= 0xff51afd7ed558ccdULL; h ^= h >> 33;
    float f = ((float)(unsigned)(h & 0xffffff) / 8388608.0f - 1.0f) * 0.03f;
    union { unsigned u; float ff; } v; v.ff = f;
    unsigned r = v.u + 0x7fff + ((v.u >> 16) & 1);
    x[i] = (unsigned short)(r >> 16);
  }
}

// V8: B-fragments read directly from global (L2-resident queries, no sB);
//     transposed epilogue. V9 = V8 + tile-max gate on the epilogue.
template <int V>
__global__ __launch_bounds__(NT, 3) void k_knn_direct(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels, int d, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  __shared__ __align__(16) char smem[48 * 1024];
  unsigned short* sA = (unsigned short*)smem;
  float* sS = (float*)smem + (8 * 1024) / 4 * 2;  // after 8KB A tile... bytes!
  sS = (float*)(smem + 8 * 1024);
  __shared__ float s_minthr[1];
  __shared__ float s_red[4];

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;

  float tv[KC];
  int ti[KC];
#pragma unroll
  for (int i = 0; i < KC; ++i) { tv[i] = -1e30f; ti[i] = -1; }
  if (tid == 0) s_minthr[0] = -1e30f;

  const long long d2 = (long long)d * 2;

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;
    float4v acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < d; kt += BK) {
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int chunk = wc * 2 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
      __syncthreads();
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = m * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>(sA + r * BK + k));
        }
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          int k = kt + ks * 32 + (lane >> 4) * 8;
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(qs + (long long)c * d + k));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[nn], acc[m][nn], 0, 0, 0);
      }
      __syncthreads();
    }

    bool do_scan = true;
    if (V == 9) {
      float vmax = -1e30f;
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int nn = 0; nn < 4; ++nn)
#pragma unroll
          for (int r = 0; r < 4; ++r) vmax = fmaxf(vmax, acc[m][nn][r]);
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        vmax = fmaxf(vmax, __shfl_xor(vmax, off, WAVE));
      if (lane == 0) s_red[wc] = vmax;
      __syncthreads();
      float tmax = fmaxf(fmaxf(s_red[0], s_red[1]), fmaxf(s_red[2], s_red[3]));
      do_scan = tmax >= s_minthr[0];
    }

    if (do_scan) {
      float newthr = 1e30f;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          int m = h * 2 + mi;
#pragma unroll
          for (int nn = 0; nn < 4; ++nn) {
            int col = wc * 64 + nn * 16 + (lane & 15);
            int s0 = mi * 16 + (lane >> 4) * 4;
            *reinterpret_cast<float4v*>(sS + col * TSTR + s0) = acc[m][nn];
          }
        }
        __syncthreads();
        const long long grow0 = prow + (long long)h * SCH;
        const float* myrow = sS + tid * TSTR;
#pragma unroll
        for (int rb = 0; rb < SCH / 4; ++rb) {
          float4v v4 = *reinterpret_cast<const float4v*>(myrow + rb * 4);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float s = v4[j];
            if (s > tv[KC - 1]) {
              float cs = s; int ci = (int)(grow0 + rb * 4 + j);
#pragma unroll
              for (int i = 0; i < KC; ++i) {
                bool ins = cs > tv[i];
                float ts2 = tv[i]; int tj = ti[i];
                tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
                cs = ins ? ts2 : cs; ci = ins ? tj : ci;
              }
            }
          }
        }
        __syncthreads();
      }
      if (V == 9) {
        // refresh block-min threshold: min over threads of tv[KC-1]
        float myth = tv[KC - 1];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
          myth = fminf(myth, __shfl_xor(myth, off, WAVE));
        if (lane == 0) s_red[wc] = myth;
        __syncthreads();
        if (tid == 0)
          s_minthr[0] = fminf(fminf(s_red[0], s_red[1]),
                              fminf(s_red[2], s_red[3]));
        __syncthreads();
      }
    }
  }

  long long slot = (long long)blockIdx.x * BN + tid;
#pragma unroll
  for (int i = 0; i < KC; ++i) {
    cand_score[slot * KC + i] = tv[i];
    cand_idx[slot * KC + i] = ti[i];
  }
}

// V10/V11: counted-vmcnt pipeline (T3/T4): raw s_barrier (no forced
// vmcnt-0 drain) + asm s_waitcnt vmcnt(N) so next-tile loads stay in
// flight across barriers. V10 = A+B double-buffered (80 KB LDS,
// 2 blocks/CU); V11 = A-only double-buffer (48 KB, 3 blocks/CU),
// B staged synchronously per K-step.
// compiler memory fence (no instruction) + HW barrier; avoids both the
// __syncthreads() forced vmcnt(0) drain AND sched_barrier(0) order-pinning
// (guide m141: order-pinning = 510 TF regression).
#define RAW_BAR() do { asm volatile("" ::: "memory"); \
                       __builtin_amdgcn_s_barrier(); \
                       asm volatile("" ::: "memory"); } while (0)
#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")

template <int V>
__global__ __launch_bounds__(NT, 3) void k_knn_pipe(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels, int d, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  constexpr bool FULL_DBUF = (V == 10);
  __shared__ __align__(16) char smem[FULL_DBUF ? 80 * 1024 : 48 * 1024];
  // layout: V10: A0 8K | B0 32K | A1 8K | B1 32K
  //         V11: A0 8K | A1 8K | B 32K
  unsigned short* A0 = (unsigned short*)smem;
  unsigned short* B0 = (unsigned short*)(smem + (FULL_DBUF ? 8192 : 16384));
  unsigned short* A1 = (unsigned short*)(smem + (FULL_DBUF ? 40960 : 8192));
  unsigned short* B1 = FULL_DBUF ? (unsigned short*)(smem + 49152) : B0;
  float* sS = (float*)smem;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;

  float tv[KC];
  int ti[KC];
#pragma unroll
  for (int i = 0; i < KC; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;
    float4v acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

#define STAGE_A(kt, dst) do { \
      _Pragma("unroll") \
      for (int it = 0; it < 2; ++it) { \
        int chunk = wc * 2 + it; \
        int byte_off = chunk * 1024 + lane * 16; \
        int r = byte_off / (BK * 2); \
        int cb = byte_off % (BK * 2); \
        const G_AS unsigned int* gp = (const G_AS unsigned int*)( \
            (const char*)db + (prow + r) * d2 + (long long)(kt) * 2 + cb); \
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)(dst) + chunk * 1024); \
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0); \
      } } while (0)
#define STAGE_B(kt, dst) do { \
      _Pragma("unroll") \
      for (int it = 0; it < 8; ++it) { \
        int chunk = wc * 8 + it; \
        int byte_off = chunk * 1024 + lane * 16; \
        int r = byte_off / (BK * 2); \
        int cb = byte_off % (BK * 2); \
        const G_AS unsigned int* gp = (const G_AS unsigned int*)( \
            (const char*)qs + (long long)r * d2 + (long long)(kt) * 2 + cb); \
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)(dst) + chunk * 1024); \
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0); \
      } } while (0)
#define MFMA_TILE(rA, rB) do { \
      _Pragma("unroll") \
      for (int ks = 0; ks < BK / 32; ++ks) { \
        bf16x8 af[4], bf[4]; \
        _Pragma("unroll") \
        for (int m = 0; m < 4; ++m) { \
          int r = m * 16 + (lane & 15); \
          int k = ks * 32 + (lane >> 4) * 8; \
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>((rA) + r * BK + k)); \
        } \
        _Pragma("unroll") \
        for (int nn = 0; nn < 4; ++nn) { \
          int c = wc * 64 + nn * 16 + (lane & 15); \
          int k = ks * 32 + (lane >> 4) * 8; \
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>((rB) + c * BK + k)); \
        } \
        _Pragma("unroll") \
        for (int m = 0; m < 4; ++m) \
          _Pragma("unroll") \
          for (int nn = 0; nn < 4; ++nn) \
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16( \
                af[m], bf[nn], acc[m][nn], 0, 0, 0); \
      } } while (0)

    const int nt = d / BK;  // assume even (d % 128 == 0)
    if (FULL_DBUF) {
      STAGE_A(0, A0); STAGE_B(0, B0);
      for (int t = 0; t < nt; t += 2) {
        if (t + 1 < nt) { STAGE_A((t + 1) * BK, A1); STAGE_B((t + 1) * BK, B1); }
        VMCNT(10);          // tile t landed; tile t+1 (10 loads) in flight
        RAW_BAR();
        MFMA_TILE(A0, B0);
        RAW_BAR();
        if (t + 2 < nt) {
          STAGE_A((t + 2) * BK, A0); STAGE_B((t + 2) * BK, B0);
          VMCNT(10);
        } else {
          VMCNT(0);
        }
        RAW_BAR();
        MFMA_TILE(A1, B1);
        RAW_BAR();
      }
    } else {
      STAGE_A(0, A0);
      for (int t = 0; t < nt; t += 2) {
        if (t + 1 < nt) STAGE_A((t + 1) * BK, A1);  // 2 loads, stays in flight
        STAGE_B(t * BK, B0);                        // 8 loads (L2-hot)
        VMCNT(2);           // drain A(t)+B(t); A(t+1)'s 2 keep flying
        RAW_BAR();
        MFMA_TILE(A0, B0);
        RAW_BAR();
        STAGE_B((t + 1) * BK, B0);
        if (t + 2 < nt) {
          STAGE_A((t + 2) * BK, A0);
          VMCNT(2);
        } else {
          VMCNT(0);
        }
        RAW_BAR();
        MFMA_TILE(A1, B0);
        RAW_BAR();
      }
    }
    VMCNT(0);
    __syncthreads();

    // transposed epilogue (V6 form)
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int m = h * 2 + mi;
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int col = wc * 64 + nn * 16 + (lane & 15);
          int s0 = mi * 16 + (lane >> 4) * 4;
          *reinterpret_cast<float4v*>(sS + col * TSTR + s0) = acc[m][nn];
        }
      }
      __syncthreads();
      const long long grow0 = prow + (long long)h * SCH;
      const float* myrow = sS + tid * TSTR;
#pragma unroll
      for (int rb = 0; rb < SCH / 4; ++rb) {
        float4v v4 = *reinterpret_cast<const float4v*>(myrow + rb * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float s = v4[j];
          if (s > tv[KC - 1]) {
            float cs = s; int ci = (int)(grow0 + rb * 4 + j);
#pragma unroll
            for (int i = 0; i < KC; ++i) {
              bool ins = cs > tv[i];
              float ts2 = tv[i]; int tj = ti[i];
              tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
              cs = ins ? ts2 : cs; ci = ins ? tj : ci;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  long long slot = (long long)blockIdx.x * BN + tid;
#pragma unroll
  for (int i = 0; i < KC; ++i) {
    cand_score[slot * KC + i] = tv[i];
    cand_idx[slot * KC + i] = ti[i];
  }
#undef STAGE_A
#undef STAGE_B
#undef MFMA_TILE
}

template <int V>
__global__ __launch_bounds__(NT, 3) void k_knn(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels, int d, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  constexpr bool DBUF = (V == 7);
  __shared__ __align__(16) char smem[DBUF ? (BM * BK + BN * BK) * 4
                                          : (BM * BK + BN * BK) * 2];
  unsigned short* sA = (unsigned short*)smem;
  unsigned short* sB = (unsigned short*)(smem + BM * BK * 2);
  float* sS = (float*)smem;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;

  float tv[KC];
  int ti[KC];
#pragma unroll
  for (int i = 0; i < KC; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;
    float4v acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    auto stage = [&](int kt, int buf) {
      unsigned short* dA = sA + buf * (BM * BK + BN * BK);
      unsigned short* dB = sB + buf * (BM * BK + BN * BK);
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int chunk = wc * 2 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)dA + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
#pragma unroll
      for (int it = 0; it < 8; ++it) {
        int chunk = wc * 8 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)dB + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
    };
    auto mfma_tile = [&](int buf) {
      const unsigned short* rA = sA + buf * (BM * BK + BN * BK);
      const unsigned short* rB = sB + buf * (BM * BK + BN * BK);
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = m * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>(rA + r * BK + k));
        }
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(rB + c * BK + k));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[nn], acc[m][nn], 0, 0, 0);
      }
    };

    if (DBUF) {
      int nt = d / BK;
      stage(0, 0);
      asm volatile("s_waitcnt vmcnt(0)");
      __syncthreads();
      int cur = 0;
      for (int t = 0; t < nt - 1; ++t) {
        stage((t + 1) * BK, cur ^ 1);
        mfma_tile(cur);
        asm volatile("s_waitcnt vmcnt(0)");
        __syncthreads();
        cur ^= 1;
      }
      mfma_tile(cur);
      __syncthreads();
    } else
    for (int kt = 0; kt < d; kt += BK) {
      if (V != 3 && V != 4) {
#pragma unroll
        for (int it = 0; it < 2; ++it) {
          int chunk = wc * 2 + it;
          int byte_off = chunk * 1024 + lane * 16;
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
      }
      if (V != 2 && V != 4) {
#pragma unroll
        for (int it = 0; it < 8; ++it) {
          int chunk = wc * 8 + it;
          int byte_off = chunk * 1024 + lane * 16;
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sB + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
      }
      __syncthreads();
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = m * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>(sA + r * BK + k));
        }
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(sB + c * BK + k));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[nn], acc[m][nn], 0, 0, 0);
      }
      __syncthreads();
    }

    if (V == 1 || V == 4) {
      // keep acc live without the epilogue
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int nn = 0; nn < 4; ++nn)
          asm volatile("" ::"v"(acc[m][nn][0]), "v"(acc[m][nn][3]));
      continue;
    }

    if (V == 6 || V == 7) {
#pragma unroll
      for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          int m = h * 2 + mi;
#pragma unroll
          for (int nn = 0; nn < 4; ++nn) {
            int col = wc * 64 + nn * 16 + (lane & 15);
            int s0 = mi * 16 + (lane >> 4) * 4;
            *reinterpret_cast<float4v*>(sS + col * TSTR + s0) = acc[m][nn];
          }
        }
        __syncthreads();
        const long long grow0 = prow + (long long)h * SCH;
        const float* myrow = sS + tid * TSTR;
#pragma unroll
        for (int rb = 0; rb < SCH / 4; ++rb) {
          float4v v4 = *reinterpret_cast<const float4v*>(myrow + rb * 4);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float s = v4[j];
            if (s > tv[KC - 1]) {
              float cs = s; int ci = (int)(grow0 + rb * 4 + j);
#pragma unroll
              for (int i = 0; i < KC; ++i) {
                bool ins = cs > tv[i];
                float ts2 = tv[i]; int tj = ti[i];
                tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
                cs = ins ? ts2 : cs; ci = ins ? tj : ci;
              }
            }
          }
        }
        __syncthreads();
      }
      long long slot0 = (long long)blockIdx.x * BN + tid;
      continue;
    }

#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int m = h * 2 + mi;
#pragma unroll
        for (int nn = 0; nn < 4; ++nn)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int srow = mi * 16 + (lane >> 4) * 4 + r;
            int col = wc * 64 + nn * 16 + (lane & 15);
            sS[srow * SSTR + col] = acc[m][nn][r];
          }
      }
      __syncthreads();
      const long long grow0 = prow + (long long)h * SCH;
      bool scan = true;
      if (V == 5) {
        // wave-cooperative chunk max: each lane maxes its column set
        float cmax = -1e30f;
        for (int r = 0; r < SCH; ++r)
          cmax = fmaxf(cmax, sS[r * SSTR + tid]);
        scan = cmax > tv[KC - 1];
        if (scan) {
#pragma unroll
          for (int r = 0; r < SCH; ++r) {
            float s = sS[r * SSTR + tid];
            if (s > tv[KC - 1]) {
              float cs = s; int ci = (int)(grow0 + r);
#pragma unroll
              for (int i = 0; i < KC; ++i) {
                bool ins = cs > tv[i];
                float ts2 = tv[i]; int tj = ti[i];
                tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
                cs = ins ? ts2 : cs; ci = ins ? tj : ci;
              }
            }
          }
        }
      } else {
#pragma unroll
        for (int r = 0; r < SCH; ++r) {
          float s = sS[r * SSTR + tid];
          if (s > tv[KC - 1]) {
            float cs = s; int ci = (int)(grow0 + r);
#pragma unroll
            for (int i = 0; i < KC; ++i) {
              bool ins = cs > tv[i];
              float ts2 = tv[i]; int tj = ti[i];
              tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
              cs = ins ? ts2 : cs; ci = ins ? tj : ci;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  long long slot = (long long)blockIdx.x * BN + tid;
#pragma unroll
  for (int i = 0; i < KC; ++i) {
    cand_score[slot * KC + i] = tv[i];
    cand_idx[slot * KC + i] = ti[i];
  }
}


// ===========================================================================
// V16: 256x256 tile, 8 waves, per-wave-PARTITIONED staging (each wave issues
// exactly the A-half + B-half it consumes), issue-early during the previous
// tile, own-loads vmcnt(0) + raw barrier once per K-tile, quadrant-phased
// MFMA with B fragments hoisted into registers per tile.
// ===========================================================================
#define BM16 256
#define TSTR16 36

__global__ __launch_bounds__(512, 2) void k_knn_v16(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels, int d, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  __shared__ __align__(16) char smem[128 * 1024];
  // layout: A0 | A1 | B0 | B1 (32 KB each); sS epilogue buffer aliases
#define V16_A(buf) ((unsigned short*)(smem + (buf) * 32768))
#define V16_B(buf) ((unsigned short*)(smem + 65536 + (buf) * 32768))
  float* sS = (float*)smem;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;       // 0..7
  const int wr = wid >> 2;        // A half consumed
  const int wc = wid & 3;
  const int bh = wc >> 1;         // B half consumed
  // staging groups: A-half wr staged by the 4 waves with this wr
  //   (group rank ga = wc in 0..3); B-half bh staged by the 4 waves with
  //   this bh (group rank gb = (wr<<1) | (wc&1)).
  const int ga = wc;
  const int gb = (wr << 1) | (wc & 1);

  float tv[KC];
  int ti[KC];
#pragma unroll
  for (int i = 0; i < KC; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM16;
    float4v acc[8][4];
#pragma unroll
    for (int m = 0; m < 8; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    // one K-tile's staging for THIS wave: 4 chunks of its A-half +
    // 4 chunks of its B-half (chunk = 64 lanes x 16 B = 1 KB).
    // A-half wr occupies sA[buf] bytes [wr*16KB, +16KB); within the half,
    // this wave covers chunks [ga*4, ga*4+4). Same for B with gb.
#define V16_STAGE_CHUNK(kt, buf, c)  do {                                   \
      int cc = (c);                                                         \
      if (cc < 4) {   /* A chunk */                                         \
        int chunk = wr * 16 + ga * 4 + cc;                                  \
        int byte_off = chunk * 1024 + lane * 16;                            \
        int r = byte_off / (BK * 2);                                        \
        int cb = byte_off % (BK * 2);                                       \
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(           \
            (const char*)db + (prow + r) * d2 + (long long)(kt) * 2 + cb);  \
        L_AS unsigned int* lp = (L_AS unsigned int*)(                       \
            (char*)V16_A(buf) + chunk * 1024);                              \
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);                 \
      } else {        /* B chunk */                                         \
        int chunk = bh * 16 + gb * 4 + (cc - 4);                            \
        int byte_off = chunk * 1024 + lane * 16;                            \
        int r = byte_off / (BK * 2);                                        \
        int cb = byte_off % (BK * 2);                                       \
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(           \
            (const char*)qs + (long long)r * d2 + (long long)(kt) * 2 + cb);\
        L_AS unsigned int* lp = (L_AS unsigned int*)(                       \
            (char*)V16_B(buf) + chunk * 1024);                              \
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);                 \
      }                                                                     \
    } while (0)

    const int nt = d / BK;
    // prologue: stage tile 0, drain, join
#pragma unroll
    for (int c = 0; c < 8; ++c) V16_STAGE_CHUNK(0, 0, c);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");

    for (int t = 0; t < nt; ++t) {
      const int cur = t & 1;
      const unsigned short* rA = V16_A(cur);
      const unsigned short* rB = V16_B(cur);
      const int lq = lane & 15;
      const int lk8 = (lane >> 4) * 8;
      // B fragments for this wave's 64 cols, whole K-tile (held in regs)
      bf16x8 bfr[4][2];
#pragma unroll
      for (int nn = 0; nn < 4; ++nn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          bfr[nn][ks] = (bf16x8)(*reinterpret_cast<const short8v*>(
              rB + (wc * 64 + nn * 16 + lq) * BK + ks * 32 + lk8));
      // 4 quadrant phases: issue 2 next-tile stages, read 2 A-frag rows,
      // 16 MFMA
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (t + 1 < nt) {
          V16_STAGE_CHUNK((t + 1) * BK, cur ^ 1, 2 * q);
          V16_STAGE_CHUNK((t + 1) * BK, cur ^ 1, 2 * q + 1);
        }
        bf16x8 af[2][2];
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            af[mi][ks] = (bf16x8)(*reinterpret_cast<const short8v*>(
                rA + (wr * 128 + q * 32 + mi * 16 + lq) * BK + ks * 32 + lk8));
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              acc[q * 2 + mi][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mi][ks], bfr[nn][ks], acc[q * 2 + mi][nn], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      // tile boundary: own next-tile loads must land; nothing newer exists
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }

    // ---- epilogue: 8 chunks of 32 rows, transposed sS [col][32+4] ----
#pragma unroll
    for (int h = 0; h < 8; ++h) {
      if (wr == (h >> 2)) {
        int q = h & 3;
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          int col = 0;
#pragma unroll
          for (int nn = 0; nn < 4; ++nn) {
            col = wc * 64 + nn * 16 + (lane & 15);
            int s0 = mi * 16 + (lane >> 4) * 4;
            *reinterpret_cast<float4v*>(sS + col * TSTR16 + s0) =
                acc[q * 2 + mi][nn];
          }
        }
      }
      __syncthreads();
      const int myq = tid & 255;
      const int half = tid >> 8;
      const long long grow0 = prow + (long long)h * 32 + half * 16;
      const float* myrow = sS + myq * TSTR16 + half * 16;
#pragma unroll
      for (int rb = 0; rb < 4; ++rb) {
        float4v v4 = *reinterpret_cast<const float4v*>(myrow + rb * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float s = v4[j];
          if (s > tv[KC - 1]) {
            float cs = s; int ci = (int)(grow0 + rb * 4 + j);
#pragma unroll
            for (int i = 0; i < KC; ++i) {
              bool ins = cs > tv[i];
              float ts2 = tv[i]; int tj = ti[i];
              tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
              cs = ins ? ts2 : cs; ci = ins ? tj : ci;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  long long slot = ((long long)blockIdx.x * 2 + (tid >> 8)) * BN + (tid & 255);
#pragma unroll
  for (int i = 0; i < KC; ++i) {
    cand_score[slot * KC + i] = tv[i];
    cand_idx[slot * KC + i] = ti[i];
  }
#undef V16_STAGE_CHUNK
#undef V16_A
#undef V16_B
}

#define KO4 10
template <int V>
__global__ __launch_bounds__(NT, 4) void k_knn_occ4(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels, int d, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  constexpr bool DBUF = (V == 7);
  __shared__ __align__(16) char smem[DBUF ? (BM * BK + BN * BK) * 4
                                          : (BM * BK + BN * BK) * 2];
  unsigned short* sA = (unsigned short*)smem;
  unsigned short* sB = (unsigned short*)(smem + BM * BK * 2);
  float* sS = (float*)smem;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;

  float tv[KO4];
  int ti[KO4];
#pragma unroll
  for (int i = 0; i < KO4; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;
    float4v acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    auto stage = [&](int kt, int buf) {
      unsigned short* dA = sA + buf * (BM * BK + BN * BK);
      unsigned short* dB = sB + buf * (BM * BK + BN * BK);
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int chunk = wc * 2 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)dA + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
#pragma unroll
      for (int it = 0; it < 8; ++it) {
        int chunk = wc * 8 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)dB + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
    };
    auto mfma_tile = [&](int buf) {
      const unsigned short* rA = sA + buf * (BM * BK + BN * BK);
      const unsigned short* rB = sB + buf * (BM * BK + BN * BK);
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = m * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>(rA + r * BK + k));
        }
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(rB + c * BK + k));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[nn], acc[m][nn], 0, 0, 0);
      }
    };

    if (DBUF) {
      int nt = d / BK;
      stage(0, 0);
      asm volatile("s_waitcnt vmcnt(0)");
      __syncthreads();
      int cur = 0;
      for (int t = 0; t < nt - 1; ++t) {
        stage((t + 1) * BK, cur ^ 1);
        mfma_tile(cur);
        asm volatile("s_waitcnt vmcnt(0)");
        __syncthreads();
        cur ^= 1;
      }
      mfma_tile(cur);
      __syncthreads();
    } else
    for (int kt = 0; kt < d; kt += BK) {
      if (V != 3 && V != 4) {
#pragma unroll
        for (int it = 0; it < 2; ++it) {
          int chunk = wc * 2 + it;
          int byte_off = chunk * 1024 + lane * 16;
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
      }
      if (V != 2 && V != 4) {
#pragma unroll
        for (int it = 0; it < 8; ++it) {
          int chunk = wc * 8 + it;
          int byte_off = chunk * 1024 + lane * 16;
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sB + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
      }
      __syncthreads();
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = m * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          af[m] = (bf16x8)(*reinterpret_cast<const short8v*>(sA + r * BK + k));
        }
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          int k = ks * 32 + (lane >> 4) * 8;
          bf[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(sB + c * BK + k));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[nn], acc[m][nn], 0, 0, 0);
      }
      __syncthreads();
    }

    if (V == 1 || V == 4) {
      // keep acc live without the epilogue
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int nn = 0; nn < 4; ++nn)
          asm volatile("" ::"v"(acc[m][nn][0]), "v"(acc[m][nn][3]));
      continue;
    }

    if (V == 6 || V == 7) {
#pragma unroll
      for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          int m = h * 2 + mi;
#pragma unroll
          for (int nn = 0; nn < 4; ++nn) {
            int col = wc * 64 + nn * 16 + (lane & 15);
            int s0 = mi * 16 + (lane >> 4) * 4;
            *reinterpret_cast<float4v*>(sS + col * TSTR + s0) = acc[m][nn];
          }
        }
        __syncthreads();
        const long long grow0 = prow + (long long)h * SCH;
        const float* myrow = sS + tid * TSTR;
#pragma unroll
        for (int rb = 0; rb < SCH / 4; ++rb) {
          float4v v4 = *reinterpret_cast<const float4v*>(myrow + rb * 4);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float s = v4[j];
            if (s > tv[KO4 - 1]) {
              float cs = s; int ci = (int)(grow0 + rb * 4 + j);
#pragma unroll
              for (int i = 0; i < KO4; ++i) {
                bool ins = cs > tv[i];
                float ts2 = tv[i]; int tj = ti[i];
                tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
                cs = ins ? ts2 : cs; ci = ins ? tj : ci;
              }
            }
          }
        }
        __syncthreads();
      }
      long long slot0 = (long long)blockIdx.x * BN + tid;
      continue;
    }

#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int m = h * 2 + mi;
#pragma unroll
        for (int nn = 0; nn < 4; ++nn)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int srow = mi * 16 + (lane >> 4) * 4 + r;
            int col = wc * 64 + nn * 16 + (lane & 15);
            sS[srow * SSTR + col] = acc[m][nn][r];
          }
      }
      __syncthreads();
      const long long grow0 = prow + (long long)h * SCH;
      bool scan = true;
      if (V == 5) {
        // wave-cooperative chunk max: each lane maxes its column set
        float cmax = -1e30f;
        for (int r = 0; r < SCH; ++r)
          cmax = fmaxf(cmax, sS[r * SSTR + tid]);
        scan = cmax > tv[KO4 - 1];
        if (scan) {
#pragma unroll
          for (int r = 0; r < SCH; ++r) {
            float s = sS[r * SSTR + tid];
            if (s > tv[KO4 - 1]) {
              float cs = s; int ci = (int)(grow0 + r);
#pragma unroll
              for (int i = 0; i < KO4; ++i) {
                bool ins = cs > tv[i];
                float ts2 = tv[i]; int tj = ti[i];
                tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
                cs = ins ? ts2 : cs; ci = ins ? tj : ci;
              }
            }
          }
        }
      } else {
#pragma unroll
        for (int r = 0; r < SCH; ++r) {
          float s = sS[r * SSTR + tid];
          if (s > tv[KO4 - 1]) {
            float cs = s; int ci = (int)(grow0 + r);
#pragma unroll
            for (int i = 0; i < KO4; ++i) {
              bool ins = cs > tv[i];
              float ts2 = tv[i]; int tj = ti[i];
              tv[i] = ins ? cs : tv[i]; ti[i] = ins ? ci : ti[i];
              cs = ins ? ts2 : cs; ci = ins ? tj : ci;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  long long slot = (long long)blockIdx.x * BN + tid;
#pragma unroll
  for (int i = 0; i < KO4; ++i) {
    cand_score[slot * KO4 + i] = tv[i];
    cand_idx[slot * KO4 + i] = ti[i];
  }
}


#undef KO4

template <int V>
float run(const unsigned short* db, const unsigned short* qs, long long n,
          int d, float* cs, int* ci, int iters) {
  long long panels = n / BM;
  int grid = (int)std::min<long long>(panels, 2048);
  // warmup
  hipLaunchKernelGGL((k_knn<V>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
  hipDeviceSynchronize();
  hipEvent_t t0, t1;
  hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL((k_knn<V>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
  hipEventRecord(t1);
  hipEventSynchronize(t1);
  float ms;
  hipEventElapsedTime(&ms, t0, t1);
  return ms / iters;
}

int main() {
  const long long n = 4 << 20;
  const int d = 1024;
  unsigned short *db, *qs;
  float* cs;
  int* ci;
  hipMalloc(&db, n * d * 2);
  hipMalloc(&qs, (long long)BN * d * 2);
  hipMalloc(&cs, 8192LL * BN * KC * 4);
  hipMalloc(&ci, 8192LL * BN * KC * 4);
  hipLaunchKernelGGL(fill_rand, dim3(4096), dim3(256), 0, 0, db, n * d);
  hipLaunchKernelGGL(fill_rand, dim3(64), dim3(256), 0, 0, qs, (long long)BN * d);
  hipDeviceSynchronize();
  double flops = 2.0 * n * d * BN;
  double bytes = (double)n * d * 2;
  const char* names[] = {"full", "no-epilogue", "no-B-stage", "no-A-stage",
                         "mfma-only", "gated-epilogue"};
  float ms;
  ms = run<0>(db, qs, n, d, cs, ci, 10);
  printf("V0 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[0], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<1>(db, qs, n, d, cs, ci, 10);
  printf("V1 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[1], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<2>(db, qs, n, d, cs, ci, 10);
  printf("V2 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[2], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<3>(db, qs, n, d, cs, ci, 10);
  printf("V3 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[3], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<4>(db, qs, n, d, cs, ci, 10);
  printf("V4 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[4], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<5>(db, qs, n, d, cs, ci, 10);
  printf("V5 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", names[5], ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<6>(db, qs, n, d, cs, ci, 10);
  printf("V6 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "transposed-epi", ms, flops/ms/1e9, bytes/ms/1e9);
  ms = run<7>(db, qs, n, d, cs, ci, 10);
  printf("V7 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "v6+dbuf-2ph", ms, flops/ms/1e9, bytes/ms/1e9);
  {
    long long panels = n / BM;
    int grid = (int)std::min<long long>(panels, 2048);
    hipLaunchKernelGGL((k_knn_direct<8>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipDeviceSynchronize();
    hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
    hipEventRecord(t0);
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL((k_knn_direct<8>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipEventRecord(t1); hipEventSynchronize(t1);
    hipEventElapsedTime(&ms, t0, t1); ms /= 10;
    printf("V8 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "B-direct-L2", ms, flops/ms/1e9, bytes/ms/1e9);
    hipLaunchKernelGGL((k_knn_direct<9>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipDeviceSynchronize();
    hipEventRecord(t0);
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL((k_knn_direct<9>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipEventRecord(t1); hipEventSynchronize(t1);
    hipEventElapsedTime(&ms, t0, t1); ms /= 10;
    printf("V9 %-15s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "V8+tilemax-gate", ms, flops/ms/1e9, bytes/ms/1e9);
    hipLaunchKernelGGL((k_knn_pipe<10>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipDeviceSynchronize();
    hipEventRecord(t0);
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL((k_knn_pipe<10>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipEventRecord(t1); hipEventSynchronize(t1);
    hipEventElapsedTime(&ms, t0, t1); ms /= 10;
    printf("V10 %-14s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "pipe-full-dbuf", ms, flops/ms/1e9, bytes/ms/1e9);
    hipLaunchKernelGGL((k_knn_pipe<11>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipDeviceSynchronize();
    hipEventRecord(t0);
    for (int i = 0; i < 10; ++i)
      hipLaunchKernelGGL((k_knn_pipe<11>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
    hipEventRecord(t1); hipEventSynchronize(t1);
    hipEventElapsedTime(&ms, t0, t1); ms /= 10;
    printf("V11 %-14s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "pipe-A-dbuf", ms, flops/ms/1e9, bytes/ms/1e9);

    {
      long long panels = n / BM;
      int grid = (int)std::min<long long>(panels, 2048);
      hipLaunchKernelGGL((k_knn_occ4<0>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
      hipDeviceSynchronize();
      hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
      hipEventRecord(t0);
      for (int i = 0; i < 10; ++i)
        hipLaunchKernelGGL((k_knn_occ4<0>), dim3(grid), dim3(NT), 0, 0, db, qs, panels, d, cs, ci);
      hipEventRecord(t1); hipEventSynchronize(t1);
      float mso; hipEventElapsedTime(&mso, t0, t1); mso /= 10;
      printf("V17 %-14s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "occ4-k10",
             mso, flops/mso/1e9, bytes/mso/1e9);
    }

    // ---- V16: refcheck at small N, then perf ----
    {
      long long nn = 4096;
      long long panels16 = nn / 256;
      int g16 = (int)panels16;
      long long slots = (long long)g16 * 2;
      hipLaunchKernelGGL(k_knn_v16, dim3(g16), dim3(512), 0, 0, db, qs,
                         panels16, d, cs, ci);
      hipDeviceSynchronize();
      hipError_t err = hipGetLastError();
      if (err != hipSuccess) { printf("V16 launch err: %s\n", hipGetErrorString(err)); return 1; }
      // host refcheck: top-1 per query vs brute force
      unsigned short* hdb = (unsigned short*)malloc(nn * d * 2);
      unsigned short* hq = (unsigned short*)malloc((long long)BN * d * 2);
      float* hcs = (float*)malloc(slots * BN * KC * 4);
      int* hci = (int*)malloc(slots * BN * KC * 4);
      hipMemcpy(hdb, db, nn * d * 2, hipMemcpyDeviceToHost);
      hipMemcpy(hq, qs, (long long)BN * d * 2, hipMemcpyDeviceToHost);
      hipMemcpy(hcs, cs, slots * BN * KC * 4, hipMemcpyDeviceToHost);
      hipMemcpy(hci, ci, slots * BN * KC * 4, hipMemcpyDeviceToHost);
      auto b2f = [](unsigned short u) { union { unsigned i; float f; } v; v.i = (unsigned)u << 16; return v.f; };
      int bad = 0;
      for (int qi = 0; qi < BN; qi += 17) {
        // brute top-1
        float best = -1e30f; long long bi = -1;
        for (long long r = 0; r < nn; ++r) {
          float acc2 = 0;
          for (int k2 = 0; k2 < d; ++k2)
            acc2 += b2f(hdb[r * d + k2]) * b2f(hq[(long long)qi * d + k2]);
          if (acc2 > best) { best = acc2; bi = r; }
        }
        // merged candidate top-1
        float gbest = -1e30f; int gi = -1;
        for (long long s2 = 0; s2 < slots; ++s2)
          for (int k2 = 0; k2 < KC; ++k2) {
            float v = hcs[(s2 * BN + qi) * KC + k2];
            if (v > gbest) { gbest = v; gi = hci[(s2 * BN + qi) * KC + k2]; }
          }
        if (gi != bi || fabsf(gbest - best) > 1e-2f * fmaxf(fabsf(best), 1.f)) {
          if (bad < 3) printf("V16 MISMATCH q=%d: got (%d, %f) want (%lld, %f)\n",
                              qi, gi, gbest, bi, best);
          bad++;
        }
      }
      printf("V16 refcheck: %s (%d bad)\n", bad ? "FAIL" : "PASS", bad);
      free(hdb); free(hq); free(hcs); free(hci);
      if (!bad) {
        long long panels = n / 256;
        int grid = (int)std::min<long long>(panels, 4096);
        hipLaunchKernelGGL(k_knn_v16, dim3(grid), dim3(512), 0, 0, db, qs, panels, d, cs, ci);
        hipDeviceSynchronize();
        hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
        hipEventRecord(t0);
        for (int i = 0; i < 10; ++i)
          hipLaunchKernelGGL(k_knn_v16, dim3(grid), dim3(512), 0, 0, db, qs, panels, d, cs, ci);
        hipEventRecord(t1); hipEventSynchronize(t1);
        float ms16; hipEventElapsedTime(&ms16, t0, t1); ms16 /= 10;
        printf("V16 %-14s %7.3f ms  %6.0f TF  %5.2f TB/s\n", "256-tile-partn",
               ms16, flops/ms16/1e9, bytes/ms16/1e9);
      }
    }
  }
  return 0;
}
