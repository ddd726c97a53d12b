// Fused MFMA score + top-k kNN kernel for large query batches.
//
// scores = DB[N,D] (bf16, row-major) x Q^T  with Q [BN,D] (bf16, row-major,
// padded to BN=256 queries), k-selection fused into the epilogue so the
// [N, Q] score matrix is never materialized in HBM. The DB shard is read
// exactly once per query batch.
//
// Structure: m97-template GEMM from the CDNA4 guide (§5) —
//   * 64x256 output tile, 4 waves (256 thr), wave-tile 64x64,
//     mfma_f32_16x16x32_bf16 fragments, BK=64 K-steps
//   * global -> LDS staging via __builtin_amdgcn_global_load_lds width 16
//   * 2-barrier K-loop; LDS score buffer UNIONed over the staging tiles
//     (40 KB/block -> 3 blocks/CU occupancy)
//   * epilogue: C chunks bounce through LDS; 1 thread per query column
//     scans rows into a private register top-K list (statically unrolled
//     insertion so it stays in VGPRs), candidates merged by k_topk_merge_i32.
//
// Replaces the reference's cublasSgemv + single-thread top-k scan
// (reference: pkg/gpu/cuda/cuda_kernels.cu:340-480).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define G_AS __attribute__((address_space(1)))
#define L_AS __attribute__((address_space(3)))

// tile geometry
#define BM 64
#define BN 256
#define BK 64
#define NTHREADS 256
#define KCAND 12
// LDS score-scan chunk: rows per chunk
#define SCH 32
#define S_STRIDE (BN + 4)

__global__ __launch_bounds__(NTHREADS, 3) void k_knn_mfma(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels,  // number of full BM-row panels
    int d,               // inner dim, % BK == 0
    long long row_base, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  // 40 KB union: K-loop uses sA (8 KB) + sB (32 KB); epilogue reuses the
  // same space as the 32x260 fp32 score chunk (33.3 KB). Barriers separate
  // the two lifetimes.
  __shared__ __align__(16) char smem[(BM * BK + BN * BK) * 2];
  unsigned short* sA = (unsigned short*)smem;
  unsigned short* sB = (unsigned short*)(smem + BM * BK * 2);
  float* sS = (float*)smem;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;  // wave col 0..3 (64 cols each); all waves row 0

  // per-thread top-K state: this thread owns query column tid.
  float tv[KCAND];
  int ti[KCAND];
#pragma unroll
  for (int i = 0; i < KCAND; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;  // row stride in bytes

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;

    float4v acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < d; kt += BK) {
      // ---- stage A (64 x 64 = 8 KB) and B (256 x 64 = 32 KB) ----
      // One global_load_lds issue = one wave writes 64 lanes x 16 B = 1 KB
      // at a wave-uniform LDS base (linear row-major tiles).
#pragma unroll
      for (int it = 0; it < 2; ++it) {  // A: 8 chunks, 4 waves x 2
        int chunk = wc * 2 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
#pragma unroll
      for (int it = 0; it < 8; ++it) {  // B: 32 chunks, 4 waves x 8
        int chunk = wc * 8 + it;
        int byte_off = chunk * 1024 + lane * 16;
        int r = byte_off / (BK * 2);
        int cb = byte_off % (BK * 2);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sB + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
      __syncthreads();

      // ---- MFMA over the staged tile: 2 k-steps of 32 ----
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

    // ---- epilogue: 2 chunks of 32 rows through LDS (aliases sA/sB) ----
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int m = h * 2 + mi;
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int srow = mi * 16 + (lane >> 4) * 4 + r;  // within 32-row chunk
            int col = wc * 64 + nn * 16 + (lane & 15);
            sS[srow * S_STRIDE + col] = acc[m][nn][r];
          }
        }
      }
      __syncthreads();
      const long long grow0 = prow + (long long)h * SCH;
#pragma unroll
      for (int r = 0; r < SCH; ++r) {
        float s = sS[r * S_STRIDE + tid];
        if (s > tv[KCAND - 1]) {
          float cs = s;
          int ci = (int)(grow0 + r);
#pragma unroll
          for (int i = 0; i < KCAND; ++i) {
            bool ins = cs > tv[i];
            float ts = tv[i]; int tj = ti[i];
            tv[i] = ins ? cs : tv[i];
            ti[i] = ins ? ci : ti[i];
            cs = ins ? ts : cs; ci = ins ? tj : ci;
          }
        }
      }
      __syncthreads();
    }
  }

  // ---- write candidates ----
  long long slot = (long long)blockIdx.x * BN + tid;
#pragma unroll
  for (int i = 0; i < KCAND; ++i) {
    cand_score[slot * KCAND + i] = tv[i];
    cand_idx[slot * KCAND + i] = ti[i];
  }
}

// Merge for int32 local indices -> int64 global (adds row_base).
template <int K>
__global__ void k_topk_merge_i32(const float* __restrict__ cand_score,
                                 const int* __restrict__ cand_idx,
                                 long long w, int q_stride, int k_out,
                                 long long row_base,
                                 float* __restrict__ out_score,
                                 long long* __restrict__ out_idx) {
  const int q = blockIdx.x;
  const long long total = w * K;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  __shared__ unsigned long long s_best[8];

  float tv[K];
  int ti[K];
#pragma unroll
  for (int i = 0; i < K; ++i) { tv[i] = -1e30f; ti[i] = -1; }
  for (long long j = threadIdx.x; j < total; j += blockDim.x) {
    long long src = ((j / K) * q_stride + q) * K + (j % K);
    float s = cand_score[src];
    if (s > tv[K - 1]) {
      int ci = cand_idx[src];
      float cs = s;
#pragma unroll
      for (int i = 0; i < K; ++i) {
        bool ins = cs > tv[i];
        float ts = tv[i]; int tj = ti[i];
        tv[i] = ins ? cs : tv[i];
        ti[i] = ins ? ci : ti[i];
        cs = ins ? ts : cs; ci = ins ? tj : ci;
      }
    }
  }

  unsigned int mono;
  int head = 0;
  for (int r = 0; r < k_out; ++r) {
    float hv = -1e30f;
#pragma unroll
    for (int i = 0; i < K; ++i)
      if (i == head) hv = tv[i];
    if (head >= K) hv = -1e30f;
    {
      unsigned int u = __float_as_uint(hv);
      mono = (u & 0x80000000u) ? ~u : (u | 0x80000000u);
    }
    unsigned long long packed =
        ((unsigned long long)mono << 32) | (unsigned int)threadIdx.x;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      unsigned long long o = __shfl_xor(packed, off, WAVE);
      if (o > packed) packed = o;
    }
    if (lane == 0) s_best[wid] = packed;
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned long long b = s_best[0];
      for (int i = 1; i < (int)(blockDim.x / WAVE); ++i)
        if (s_best[i] > b) b = s_best[i];
      s_best[0] = b;
    }
    __syncthreads();
    int winner = (int)(s_best[0] & 0xffffffffu);
    if (threadIdx.x == winner) {
      float wv = -1e30f; int wi = -1;
#pragma unroll
      for (int i = 0; i < K; ++i)
        if (i == head) { wv = tv[i]; wi = ti[i]; }
      out_score[(long long)q * k_out + r] = wv;
      out_idx[(long long)q * k_out + r] = (wi < 0) ? -1 : row_base + wi;
      head++;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// host wrapper: full-panel part of the shard only (n_panels * BM rows).
// The python side handles the tail rows and q-padding to 256.
// ---------------------------------------------------------------------------
std::tuple<at::Tensor, at::Tensor> knn_mfma(at::Tensor db, at::Tensor q,
                                            long long row_base, int k_out) {
  TORCH_CHECK(db.is_cuda() && db.dim() == 2 && db.is_contiguous() &&
                  db.scalar_type() == at::kBFloat16,
              "knn_mfma: db must be contiguous 2D bf16 CUDA");
  TORCH_CHECK(q.is_cuda() && q.dim() == 2 && q.is_contiguous() &&
                  q.scalar_type() == at::kBFloat16,
              "knn_mfma: q must be contiguous 2D bf16 CUDA");
  long long n = db.size(0);
  int d = (int)db.size(1);
  TORCH_CHECK(q.size(0) == BN, "knn_mfma: q must be padded to ", BN, " rows");
  TORCH_CHECK(q.size(1) == d, "dim mismatch");
  TORCH_CHECK(d % BK == 0, "knn_mfma needs D % 64 == 0");
  TORCH_CHECK(n % BM == 0, "knn_mfma needs N % ", BM, " == 0 (python pads/tails)");
  TORCH_CHECK(n < (1LL << 31), "shard too large for int32 local rows");
  TORCH_CHECK(k_out >= 1 && k_out <= KCAND);

  long long n_panels = n / BM;
  // Enough blocks that each retires quickly (load balance + lets other
  // streams' kernels co-schedule), few enough that the candidate buffer
  // and merge stay small.
  int grid = (int)std::min<long long>(n_panels, 8192);
  auto stream = at::hip::getCurrentHIPStream().stream();

  auto opts_f = db.options().dtype(at::kFloat);
  auto opts_i32 = db.options().dtype(at::kInt);
  auto opts_i64 = db.options().dtype(at::kLong);
  at::Tensor cand_s = at::empty({grid, BN, KCAND}, opts_f);
  at::Tensor cand_i = at::empty({grid, BN, KCAND}, opts_i32);

  hipLaunchKernelGGL(k_knn_mfma, dim3(grid), dim3(NTHREADS), 0, stream,
                     (const unsigned short*)db.data_ptr(),
                     (const unsigned short*)q.data_ptr(), n_panels, d,
                     row_base, cand_s.data_ptr<float>(),
                     cand_i.data_ptr<int>());
  HIP_CHECK_LAST();

  at::Tensor out_s = at::empty({BN, k_out}, opts_f);
  at::Tensor out_i = at::empty({BN, k_out}, opts_i64);
  hipLaunchKernelGGL((k_topk_merge_i32<KCAND>), dim3(BN), dim3(256), 0, stream,
                     cand_s.data_ptr<float>(), cand_i.data_ptr<int>(), grid,
                     BN, k_out, row_base, out_s.data_ptr<float>(),
                     reinterpret_cast<long long*>(out_i.data_ptr<int64_t>()));
  HIP_CHECK_LAST();
  return {out_s, out_i};
}
