// Fused MFMA score + top-k kNN kernel for large query batches.
//
// scores = DB[N,D] (bf16, row-major) x Q^T  with Q [BN,D] (bf16, row-major,
// padded to BN=256 queries), k-selection fused into the epilogue so the
// [N, Q] score matrix is never materialized in HBM. The DB shard is read
// exactly once per query batch.
//
// Structure: m97-template GEMM from the CDNA4 guide (§5) —
//   * 128x256 output tile, 8 waves (512 thr), wave-tile 64x64,
//     mfma_f32_16x16x32_bf16 fragments, BK=64 K-steps
//   * global -> LDS staging via __builtin_amdgcn_global_load_lds width 16
//   * 2-barrier K-loop (single-buffered LDS)
//   * epilogue: C chunks bounce through LDS; 2 threads per query column
//     scan rows into private register top-K lists (statically unrolled
//     insertion so they stay in VGPRs), candidates merged by k_topk_merge.
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
#define BM 128
#define BN 256
#define BK 64
#define NTHREADS 512
#define KCAND 16
// LDS score-scan chunk: rows per chunk
#define SCH 32
#define S_STRIDE (BN + 4)

__global__ __launch_bounds__(NTHREADS, 1) void k_knn_mfma(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels,  // number of full BM-row panels
    int d,               // inner dim, % BK == 0
    long long row_base, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  __shared__ unsigned short sA[BM * BK];        // 16 KB
  __shared__ unsigned short sB[BN * BK];        // 32 KB
  __shared__ float sS[SCH * S_STRIDE];          // 33 KB score chunk

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;        // 0..7
  const int wr = wid >> 2;           // wave row 0..1  (64 rows each)
  const int wc = wid & 3;            // wave col 0..3  (64 cols each)

  // per-thread top-K state: this thread owns query column (tid & 255),
  // row-half (tid >> 8) of each score chunk.
  const int own_q = tid & (BN - 1);
  const int own_half = tid >> 8;  // 0 or 1
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
      // ---- stage A (BM x BK = 16 KB) and B (BN x BK = 32 KB) ----
      // Each global_load_lds issue: one wave writes 64 lanes x 16 B = 1 KB,
      // LDS destination is wave-uniform base + lane*16 (linear layout).
      {
        // A: 16 chunks of 1 KB; 8 waves x 2 issues
#pragma unroll
        for (int it = 0; it < 2; ++it) {
          int chunk = wid * 2 + it;
          int byte_off = chunk * 1024 + lane * 16;   // within tile
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)db + (prow + r) * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)(
              (char*)sA + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
        // B: 32 chunks of 1 KB; 8 waves x 4 issues
#pragma unroll
        for (int it = 0; it < 4; ++it) {
          int chunk = wid * 4 + it;
          int byte_off = chunk * 1024 + lane * 16;
          int r = byte_off / (BK * 2);
          int cb = byte_off % (BK * 2);
          const G_AS unsigned int* gp = (const G_AS unsigned int*)(
              (const char*)qs + (long long)r * d2 + (long long)kt * 2 + cb);
          L_AS unsigned int* lp = (L_AS unsigned int*)(
              (char*)sB + chunk * 1024);
          __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
      }
      asm volatile("s_waitcnt vmcnt(0)");
      __syncthreads();

      // ---- MFMA over the staged tile: 2 k-steps of 32 ----
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        // A fragment: row = wr*64 + m*16 + (lane&15), k = (lane>>4)*8 + ks*32
        bf16x8 af[4], bf[4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          int r = wr * 64 + m * 16 + (lane & 15);
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

    // ---- epilogue: 4 chunks of 32 rows bounce through LDS, then scan ----
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      // waves with wr == h/2 own these rows; fragment m = (h&1)*2 + {0,1}
      if (wr == (h >> 1)) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          int m = (h & 1) * 2 + mi;
#pragma unroll
          for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              int row = m * 16 + (lane >> 4) * 4 + r;   // within 64-row wave tile
              int srow = row - (h & 1) * 32;            // within 32-row chunk
              int col = wc * 64 + nn * 16 + (lane & 15);
              sS[srow * S_STRIDE + col] = acc[m][nn][r];
            }
          }
        }
      }
      __syncthreads();
      // scan: thread owns column own_q, rows [own_half*16, +16)
      const long long grow0 = prow + (long long)h * SCH + own_half * 16;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float s = sS[(own_half * 16 + r) * S_STRIDE + own_q];
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

  // ---- write candidates: slot = (blockIdx*2 + half) ----
  long long slot = ((long long)blockIdx.x * 2 + own_half) * BN + own_q;
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
// host wrapper: full-panel part of the shard only (n_panels * 128 rows).
// The python side handles the <128-row tail and q-padding to 256.
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
  TORCH_CHECK(n % BM == 0, "knn_mfma needs N % 128 == 0 (python pads/tails)");
  TORCH_CHECK(n / BM < (1LL << 31), "shard too large for int32 local rows");
  TORCH_CHECK(k_out >= 1 && k_out <= KCAND);

  long long n_panels = n / BM;
  int grid = (int)std::min<long long>(n_panels, 1024);
  auto stream = at::hip::getCurrentHIPStream().stream();

  auto opts_f = db.options().dtype(at::kFloat);
  auto opts_i32 = db.options().dtype(at::kInt);
  auto opts_i64 = db.options().dtype(at::kLong);
  long long slots = (long long)grid * 2;
  at::Tensor cand_s = at::empty({slots, BN, KCAND}, opts_f);
  at::Tensor cand_i = at::empty({slots, BN, KCAND}, opts_i32);

  hipLaunchKernelGGL(k_knn_mfma, dim3(grid), dim3(NTHREADS), 0, stream,
                     (const unsigned short*)db.data_ptr(),
                     (const unsigned short*)q.data_ptr(), n_panels, d,
                     row_base, cand_s.data_ptr<float>(),
                     cand_i.data_ptr<int>());
  HIP_CHECK_LAST();

  at::Tensor out_s = at::empty({BN, k_out}, opts_f);
  at::Tensor out_i = at::empty({BN, k_out}, opts_i64);
  hipLaunchKernelGGL((k_topk_merge_i32<KCAND>), dim3(BN), dim3(256), 0, stream,
                     cand_s.data_ptr<float>(), cand_i.data_ptr<int>(), slots,
                     BN, k_out, row_base, out_s.data_ptr<float>(),
                     reinterpret_cast<long long*>(out_i.data_ptr<int64_t>()));
  HIP_CHECK_LAST();
  return {out_s, out_i};
}
