// Standalone A/B: FA_PSTRIDE/FA_VSTRIDE padding variants for
// k_flash_attn_nc (copy of the production kernel parameterized by
// strides; refcheck vs itself at default).
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

#define FA_D 64
#define FA_KT 64
#define FA_QT 64

__device__ __forceinline__ float b2f(unsigned short u) {
  union { unsigned i; float f; } v; v.i = (unsigned)u << 16; return v.f;
}
__device__ __forceinline__ unsigned short f2b(float f) {
  union { unsigned i; float f2; } v; v.f2 = f;
  unsigned r = v.i + 0x7fff + ((v.i >> 16) & 1);
  return (unsigned short)(r >> 16);
}

template <int PSTR, int VSTR, int VMODE = 0>
__global__ __launch_bounds__(256, 2) void k_fa(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    int n_heads, int s, float scale,
    long long q_bs, long long q_ss, long long k_bs, long long k_ss,
    long long v_bs, long long v_ss) {
  __shared__ unsigned short sK[FA_KT * FA_D];
  __shared__ unsigned short sVT[FA_D * VSTR];
  __shared__ unsigned short sP[4][16 * PSTR];

  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int b = blockIdx.x / n_heads;
  const int h = blockIdx.x % n_heads;
  const int q0 = blockIdx.y * FA_QT;
  const long long hoff = (long long)h * FA_D;
  const int lq = lane & 15;
  const int lk8 = (lane >> 4) * 8;

  bf16x8 qf[2];
  {
    const unsigned short* qp =
        Q + b * q_bs + (long long)(q0 + wid * 16 + lq) * q_ss + hoff;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      qf[ks] = (bf16x8)(*reinterpret_cast<const short8v*>(qp + ks * 32 + lk8));
  }
  float m_run[4], l_run[4];
  float4v acc[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int nn = 0; nn < 4; ++nn) acc[nn] = {0.f, 0.f, 0.f, 0.f};
  const float log2e = 1.44269504f;

  for (int k0 = 0; k0 < s; k0 += FA_KT) {
    {
      const unsigned short* kp = K + b * k_bs + hoff;
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        int chunk = wid * 2 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (FA_D * 2);
        int cb = byte_off % (FA_D * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)(kp + (long long)(k0 + r) * k_ss) + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sK + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
    }
    if (VMODE == 0) {
      const unsigned short* vp = V + b * v_bs + hoff;
      int kk = threadIdx.x & 63;
      int db = (threadIdx.x >> 6) * 16;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        int dd = db + half * 8;
        short8v v = *reinterpret_cast<const short8v*>(
            vp + (long long)(k0 + kk) * v_ss + dd);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          sVT[(dd + j) * VSTR + kk] = (unsigned short)v[j];
      }
    } else {
      // VMODE 1: gather 8 consecutive k at fixed d (strided global reads,
      // V tile L1/L2-resident) -> ONE b128 LDS write per (d, k8)
      const unsigned short* vp = V + b * v_bs + hoff;
      int dd = threadIdx.x & 63;          // d fixed per thread
      int k8 = (threadIdx.x >> 6) * 16;   // two k-octets per thread
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        int kk = k8 + half * 8;
        short8v o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = (short)vp[(long long)(k0 + kk + j) * v_ss + dd];
        *reinterpret_cast<short8v*>(sVT + dd * VSTR + kk) = o;
      }
    }
    __syncthreads();

    float4v sfrag[4];
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {
      float4v c4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 bf = (bf16x8)(*reinterpret_cast<const short8v*>(
            sK + (nn * 16 + lq) * FA_D + ks * 32 + lk8));
        c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], bf, c4, 0, 0, 0);
      }
      sfrag[nn] = c4;
    }
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -1e30f;
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) mx = fmaxf(mx, sfrag[nn][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
      pmax[r] = mx * scale;
    }
    float rescale[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mn = fmaxf(m_run[r], pmax[r]);
      rescale[r] = exp2f((m_run[r] - mn) * log2e);
      m_run[r] = mn;
      l_run[r] *= rescale[r];
    }
    float lsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f((sfrag[nn][r] * scale - m_run[r]) * log2e);
        lsum[r] += p;
        sP[wid][((lane >> 4) * 4 + r) * PSTR + nn * 16 + lq] = f2b(p);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float ls = lsum[r];
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        ls += __shfl_xor(ls, off, WAVE);
      l_run[r] += ls;
    }
#pragma unroll
    for (int nn = 0; nn < 4; ++nn)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[nn][r] *= rescale[r];
    __syncthreads();
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {
      float4v c4 = acc[nn];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 af = (bf16x8)(*reinterpret_cast<const short8v*>(
            sP[wid] + lq * PSTR + ks * 32 + lk8));
        bf16x8 bf = (bf16x8)(*reinterpret_cast<const short8v*>(
            sVT + (nn * 16 + lq) * VSTR + ks * 32 + lk8));
        c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c4, 0, 0, 0);
      }
      acc[nn] = c4;
    }
    __syncthreads();
  }
  const long long o_ss = (long long)n_heads * FA_D;
  const long long obase = ((long long)b * s + q0 + wid * 16) * o_ss + hoff;
#pragma unroll
  for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      float v = acc[nn][r] / fmaxf(l_run[r], 1e-20f);
      O[obase + row * o_ss + nn * 16 + lq] = f2b(v);
    }
  }
}

__global__ void fill_rand(unsigned short* x, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long st = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += st) {
    unsigned long long h = i * 0x9e3779b97f4a7c15ULL;
    h ^= h >> 33; h *= 0xff51afd7ed558ccdULL; h ^= h >> 33;
    float f = ((float)(unsigned)(h & 0xffffff) / 8388608.0f - 1.0f) * 0.3f;
    union { unsigned u; float ff; } v; v.ff = f;
    unsigned r = v.u + 0x7fff + ((v.u >> 16) & 1);
    x[i] = (unsigned short)(r >> 16);
  }
}

template <int PSTR, int VSTR, int VMODE = 0>
float run(const unsigned short* Q, const unsigned short* K,
          const unsigned short* V, unsigned short* O,
          int B, int H, int S, int iters) {
  dim3 grid(B * H, S / FA_QT);
  long long ss = (long long)H * FA_D;
  long long bs = (long long)S * ss;
  float scale = 1.0f / sqrtf((float)FA_D);
  auto launch = [&]() {
    hipLaunchKernelGGL((k_fa<PSTR, VSTR, VMODE>), grid, dim3(256), 0, 0,
                       Q, K, V, O, H, S, scale, bs, ss, bs, ss, bs, ss);
  };
  launch();
  hipDeviceSynchronize();
  hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
  hipEventRecord(t0);
  for (int i = 0; i < iters; ++i) launch();
  hipEventRecord(t1); hipEventSynchronize(t1);
  float ms; hipEventElapsedTime(&ms, t0, t1);
  return ms / iters;
}

int main() {
  const int B = 256, H = 16, S = 256;
  long long n = (long long)B * S * H * FA_D;
  unsigned short *Q, *K, *V, *O, *O2;
  hipMalloc(&Q, n * 2); hipMalloc(&K, n * 2); hipMalloc(&V, n * 2);
  hipMalloc(&O, n * 2); hipMalloc(&O2, n * 2);
  hipLaunchKernelGGL(fill_rand, dim3(2048), dim3(256), 0, 0, Q, n);
  hipLaunchKernelGGL(fill_rand, dim3(2048), dim3(256), 0, 0, K, n);
  hipLaunchKernelGGL(fill_rand, dim3(2048), dim3(256), 0, 0, V, n);
  hipDeviceSynchronize();
  double fl = 4.0 * B * H * (double)S * S * FA_D;
  float ms;
  ms = run<72, 72>(Q, K, V, O, B, H, S, 20);
  printf("P72/V72 (prod): %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<68, 68>(Q, K, V, O2, B, H, S, 20);
  printf("P68/V68:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  // refcheck 68 vs 72
  {
    unsigned short* h1 = (unsigned short*)malloc(n * 2);
    unsigned short* h2 = (unsigned short*)malloc(n * 2);
    hipMemcpy(h1, O, n * 2, hipMemcpyDeviceToHost);
    hipMemcpy(h2, O2, n * 2, hipMemcpyDeviceToHost);
    long long bad = 0;
    for (long long i = 0; i < n; i += 997)
      if (h1[i] != h2[i]) bad++;
    printf("refcheck 68vs72: %s\n", bad ? "FAIL" : "PASS");
    free(h1); free(h2);
  }
  ms = run<80, 80>(Q, K, V, O2, B, H, S, 20);
  printf("P80/V80:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<66, 66>(Q, K, V, O2, B, H, S, 20);
  printf("P66/V66:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<72, 68>(Q, K, V, O2, B, H, S, 20);
  printf("P72/V68:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<68, 72>(Q, K, V, O2, B, H, S, 20);
  printf("P68/V72:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 72>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V72:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<68, 76>(Q, K, V, O2, B, H, S, 20);
  printf("P68/V76:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<66, 72>(Q, K, V, O2, B, H, S, 20);
  printf("P66/V72:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<68, 66>(Q, K, V, O2, B, H, S, 20);
  printf("P68/V66:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<76, 72>(Q, K, V, O2, B, H, S, 20);
  printf("P76/V72:        %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 72, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V72/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 66, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V66/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 68, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V68/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 76, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V76/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 80, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V80/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  ms = run<64, 96, 1>(Q, K, V, O2, B, H, S, 20);
  printf("P64/V96/gatherV: %6.3f ms  %5.0f TF\n", ms, fl / ms / 1e9);
  {
    run<64, 72, 0>(Q, K, V, O, B, H, S, 1);
    run<64, 72, 1>(Q, K, V, O2, B, H, S, 1);
    unsigned short* h1 = (unsigned short*)malloc(n * 2);
    unsigned short* h2 = (unsigned short*)malloc(n * 2);
    hipMemcpy(h1, O, n * 2, hipMemcpyDeviceToHost);
    hipMemcpy(h2, O2, n * 2, hipMemcpyDeviceToHost);
    long long bad = 0;
    for (long long i = 0; i < n; i += 97)
      if (h1[i] != h2[i]) bad++;
    printf("refcheck gatherV vs scalarV: %s\n", bad ? "FAIL" : "PASS");
    free(h1); free(h2);
  }
  return 0;
}
