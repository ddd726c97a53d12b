// Fused MFMA score + top-k kNN kernel for large query batches.
//
// scores = DB[N,D] (bf16, row-major) x Q^T  with Q [BN,D] (bf16, row-major,
// padded to BN=256 queries), k-selection fused into the epilogue so the
// [N, Q] score matrix is never materialized in HBM. The DB shard is read
// exactly once per query batch.
//
// Structure (chosen by ablation, scripts/knn8p.hip — see profiles/README.md):
//   * 96x256 output tile, 4 waves (256 thr), 3 workgroups/CU. Occupancy is
//     the LDS-DMA throughput lever on gfx950: global_load_lds ingest
//     scales with resident waves (1-WG/8-wave kernels cap at ~3 TB/s
//     aggregate DMA; 12 waves across 3 WGs reach ~8.8 TB/s), so the
//     "big-tile 1-WG pipelined" template loses to 3 small co-resident WGs.
//   * row-XOR LDS swizzle: 16-B slot index XORed with (row & 7) inside
//     each 128-B row; kills the 16-way ds_read bank conflict of
//     16-consecutive-rows-at-one-column fragment reads (measured:
//     SQ_LDS_BANK_CONFLICT -> 0, +9% kernel).
//   * mfma_f32_16x16x32_bf16 fragments, BK=64 K-steps, staging via
//     __builtin_amdgcn_global_load_lds width 16.
//   * transposed epilogue: acc fragments land in a [col][36] fp32 LDS
//     chunk with float4 stores/scans (4 consecutive rows per C-register
//     quad); each thread owns one query column and keeps a register
//     top-K list (statically unrolled insertion).
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
#ifndef KNN_BM
#define KNN_BM 96
#endif
#define BM KNN_BM
#define BN 256
#define BK 64
#define NTHREADS 256
#define KCAND 10
#define MW (BM / 16)

// row-XOR swizzle (self-inverse): within a 128 B row, XOR the 16 B-slot
// index with (row & 7).
__device__ __forceinline__ int swz(int b) {
  return (b & ~127) | ((b & 127) ^ (((b >> 7) & 7) << 4));
}

__global__ __launch_bounds__(NTHREADS, 3) void k_knn_mfma(
    const unsigned short* __restrict__ db, const unsigned short* __restrict__ qs,
    long long n_panels,  // number of full BM-row panels
    int d,               // inner dim, % BK == 0
    long long row_base, float* __restrict__ cand_score,
    int* __restrict__ cand_idx) {
  // 44 KB: K-loop uses sA (12 KB) + sB (32 KB); epilogue reuses the same
  // space as the [256][36] fp32 transposed chunk (36.9 KB).
  __shared__ __align__(16) char smem[BM * BK * 2 + BN * BK * 2];
  unsigned short* sA = (unsigned short*)smem;
  unsigned short* sB = (unsigned short*)(smem + BM * BK * 2);
  float* sT = (float*)smem;  // epilogue alias [256][36]

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;  // wave col 0..3 (64 cols each)

  // per-thread top-K state: this thread owns query column tid.
  float tv[KCAND];
  int ti[KCAND];
#pragma unroll
  for (int i = 0; i < KCAND; ++i) { tv[i] = -1e30f; ti[i] = -1; }

  const long long d2 = (long long)d * 2;  // row stride in bytes

  for (long long panel = blockIdx.x; panel < n_panels; panel += gridDim.x) {
    const long long prow = panel * BM;

    float4v acc[MW][4];
#pragma unroll
    for (int m = 0; m < MW; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) acc[m][nn] = {0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < d; kt += BK) {
      // ---- stage A (96x64 = 12 KB) and B (256x64 = 32 KB) ----
      // One global_load_lds = one wave writes 64 lanes x 16 B = 1 KB at a
      // wave-uniform LDS base; the global source address is pre-swizzled
      // so the LDS image is the swizzled layout.
#pragma unroll
      for (int it = 0; it < BM / 32; ++it) {  // A: BM/8 chunks, 4 waves
        int chunk = wc * (BM / 32) + it;
        int x = chunk * 1024 + lane * 16;
        int p = swz(x);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)db + (prow + (p >> 7)) * d2 + (long long)kt * 2 +
            (p & 127));
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
#pragma unroll
      for (int it = 0; it < 8; ++it) {  // B: 32 chunks, 4 waves x 8
        int chunk = wc * 8 + it;
        int x = chunk * 1024 + lane * 16;
        int p = swz(x);
        const G_AS unsigned int* gp = (const G_AS unsigned int*)(
            (const char*)qs + (long long)(p >> 7) * d2 + (long long)kt * 2 +
            (p & 127));
        L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sB + chunk * 1024);
        __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
      }
      __syncthreads();

      // ---- MFMA over the staged tile: 2 k-steps of 32 ----
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int kb = (ks * 32 + (lane >> 4) * 8) * 2;
        bf16x8 bfr[4];
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int c = wc * 64 + nn * 16 + (lane & 15);
          bfr[nn] = (bf16x8)(*reinterpret_cast<const short8v*>(
              (const char*)sB + swz(c * 128 + kb)));
        }
#pragma unroll
        for (int m = 0; m < MW; ++m) {
          int r = m * 16 + (lane & 15);
          bf16x8 af = (bf16x8)(*reinterpret_cast<const short8v*>(
              (const char*)sA + swz(r * 128 + kb)));
#pragma unroll
          for (int nn = 0; nn < 4; ++nn)
            acc[m][nn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af, bfr[nn], acc[m][nn], 0, 0, 0);
        }
      }
      __syncthreads();
    }

    // ---- epilogue: 32-row chunks through the transposed LDS block ----
#pragma unroll
    for (int h = 0; h < BM / 32; ++h) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int m = h * 2 + mi;
#pragma unroll
        for (int nn = 0; nn < 4; ++nn) {
          int col = wc * 64 + nn * 16 + (lane & 15);
          int srow = mi * 16 + (lane >> 4) * 4;
          *reinterpret_cast<float4v*>(sT + col * 36 + srow) = acc[m][nn];
        }
      }
      __syncthreads();
      const long long grow0 = prow + (long long)h * 32;
#pragma unroll
      for (int j = 0; j < 32; j += 4) {
        float4v v = *reinterpret_cast<const float4v*>(sT + tid * 36 + j);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          float s = v[e];
          if (s > tv[KCAND - 1]) {
            float cs = s;
            int ci = (int)(grow0 + j + e);
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
  // and merge stay small. At grid 8192 a 100M-corpus block runs ~6 ms,
  // which blocks embed-stream GEMMs from co-scheduling; tunable for the
  // overlap experiment via NORNICDB_KNN_GRID.
  // multiple of the resident-WG count (3 WGs/CU x 256 CUs = 768) so
  // every dispatch wave is full: at 8192 the 11th wave ran 512 WGs with
  // 2/3 of the chip idle (measured ~2-3% of the kernel)
  int max_grid = 7680;
  if (const char* g = getenv("NORNICDB_KNN_GRID")) max_grid = atoi(g);
  int grid = (int)std::min<long long>(n_panels, max_grid);
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

long long knn_bm_value() { return BM; }
