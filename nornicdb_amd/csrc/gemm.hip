// Hand-written CDNA4 MFMA GEMM for the bge-m3 encoder hot path.
//
// C[M,N] = A[M,K] x W[N,K]^T (+ bias[N]) (+ GELU), all bf16 in / bf16 out,
// fp32 accumulate. W is the torch nn.Linear weight layout [out, in] so both
// operands are K-major and stage identically.
//
// Replaces hipBLASLt (Tensile Cijk_* kernels) for the encoder QKV / attn-out /
// FFN GEMMs, and replaces the reference's cublas GEMV path + llama.cpp GEMMs
// (reference: pkg/gpu/cuda/cuda_kernels.cu:340-375, pkg/localllm/llama.go).
//
// Structure: 256x256 output tile, BK=64, 8 waves (512 thr) in a 2x4 wave
// grid (each wave owns a 128x64 sub-tile, acc[8][4] fp32x4 fragments),
// deep-pipelined 8-phase K-loop over 2 K-tiles per iteration:
//   - LDS double buffer: sA[2][256][64] + sB[2][256][64] bf16 = 128 KiB,
//     dbuf = K-tile parity. Staged with __builtin_amdgcn_global_load_lds
//     width 16 (wave-uniform LDS base + pre-swizzled global source).
//   - row-XOR swizzle: 16B slot index XORed with ((row >> 1) & 7) inside
//     each 128 B row. Fragment reads touch 16 consecutive rows at one
//     slot; with the (row & 7) variant (knn_mfma.hip) rows r and r+8
//     still collide mod 256 B (same XOR, same row parity) leaving a
//     2-way conflict — (row>>1)&7 pairs with the row-parity bit to give
//     16 distinct slots mod 256 B: conflict-free.
//   - per phase: ds_read fragments -> issue half-tile prefetch ->
//     s_barrier -> setprio(1) -> 16 x mfma_f32_16x16x32_bf16 ->
//     setprio(0) -> s_barrier (two barriers per phase; the one-barrier
//     variant measured 3-9% SLOWER — wave skew starves the staged-tile
//     cadence). Raw s_barrier (NOT __syncthreads) so prefetch loads
//     stay in flight across phase boundaries; the only vmcnt waits are
//     s_waitcnt vmcnt(8) before the group-crossing barriers (end of
//     phases 3 and 7), each gating the half-tiles staged 4-5 phases
//     earlier while leaving the most recent 8 loads in flight.
//   Safety argument for the raw barriers (audited, see repo NOTES.md):
//     * a slot is staged only in a phase strictly after its last ds_read
//       phase within the group (B slots: read ph 0,1, staged ph 2; A
//       slots: read ph 0,2, staged ph 3), and every phase's ds_reads are
//       consumed by that phase's MFMAs, so a wave reaching the end
//       barrier has its reads complete in registers.
//     * cross-wave DMA visibility: each wave's vmcnt only tracks its own
//       stages, so the vmcnt(8) is placed BEFORE an s_barrier - after
//       that barrier every wave has retired its own stage instructions
//       for the gated slots, hence all chunks of the slot are in LDS.
//   - epilogue: fp32 acc + bias (+erf-GELU) -> bf16 through a per-wave
//     16 KiB LDS bounce so the global C store is 16 B coalesced.
//
// Perf target: >= hipBLASLt's measured ~1.3 PF on the encoder shapes
// (M=65536, N in {1024,3072,4096}, K in {1024,4096}); the template's
// measured band elsewhere is 1.5-1.7 PF bf16.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define G_AS __attribute__((address_space(1)))
#define L_AS __attribute__((address_space(3)))

#define GM_BM 256
#define GM_BN 256
#define GM_BK 64
#define GM_NTHREADS 512

namespace gemm_nt_detail {

// row-XOR swizzle, involution: within each 128 B row, XOR the 16 B slot
// index with ((row >> 1) & 7) — see header: conflict-free for 16-row
// column-fragment reads (the XOR value lives in bits 8-10, untouched by
// the XOR itself, so swz(swz(x)) == x).
__device__ __forceinline__ int swz(int b) {
  return (b & ~127) | ((b & 127) ^ (((b >> 8) & 7) << 4));
}

// Stage one 1 KB wave-chunk of a [rows][GM_BK] bf16 K-major tile.
// `gsrc` is the (wave-uniform) tile base, `soff` the precomputed per-lane
// swizzled byte offset within the tile (32-bit: tiles are < 2 MB), so the
// address is saddr + voffset and costs one VGPR, not a 64-bit chain.
// chunk c covers linear LDS bytes [c*1024, (c+1)*1024); the global source
// is pre-swizzled so the LDS image is the swizzled layout.
__device__ __forceinline__ void stage_chunk(const char* __restrict__ gsrc,
                                            int soff, char* lds_base,
                                            int chunk) {
  const G_AS unsigned int* gp = (const G_AS unsigned int*)(gsrc + soff);
  L_AS unsigned int* lp = (L_AS unsigned int*)(lds_base + chunk * 1024);
  __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
}

__device__ __forceinline__ float gelu_erf(float f) {
  return 0.5f * f * (1.0f + erff(f * 0.70710678f));
}

}  // namespace gemm_nt_detail

// ACT: 0 = none, 1 = erf GELU. HAS_BIAS toggles the bias[N] add.
// ABL bits (ablation probe, perf diagnosis only — results are wrong when
// nonzero): 1 = skip in-loop staging, 2 = skip ds_reads (+waits),
// 4 = skip MFMA (frags kept alive via empty asm, guide rule 17),
// 8 = skip the global C store.
template <int ACT, bool HAS_BIAS, int ABL = 0>
__global__ __launch_bounds__(GM_NTHREADS, 1) void k_gemm_nt(
    const unsigned short* __restrict__ A,  // [M,K] bf16 row-major
    const unsigned short* __restrict__ W,  // [N,K] bf16 row-major
    const unsigned short* __restrict__ bias,  // [N] bf16 (or nullptr)
    unsigned short* __restrict__ C,        // [M,N] bf16 row-major
    int tiles_m, int tiles_n, long long K, long long M_stride_unused,
    long long N) {
  using namespace gemm_nt_detail;
  // 128 KiB: sA[2][256][64] + sB[2][256][64] bf16.
  __shared__ __align__(16) char smem[4 * GM_BM * GM_BK * 2];
  char* sA[2] = {smem, smem + GM_BM * GM_BK * 2};
  char* sB[2] = {smem + 2 * GM_BM * GM_BK * 2, smem + 3 * GM_BM * GM_BK * 2};

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;          // 0..7
  const int wr = wid >> 2;             // wave row 0..1 (128 rows each)
  const int wcol = wid & 3;            // wave col 0..3 (64 cols each)

  // XCD-aware bijective block remap (8 XCDs round-robin hardware dispatch;
  // make consecutive remapped ids land on one XCD so B-tiles share its L2).
  int nwg = tiles_m * tiles_n;
  int wg = blockIdx.x;
  {
    // bijective for any nwg: xcd < r spans get q+1 slots, the rest q.
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wg & 7, slot = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  }
  // Band-major tile walk: bands of 8 tm-panels sweep all tn before
  // advancing, so a band's A panels (8 x 512 KB at K=1024) stay resident
  // in L2/LLC across the whole tn sweep instead of being re-fetched from
  // HBM once per tn group. Within a band tm varies fastest, so the ~32
  // co-resident blocks of one XCD share a single W tile in its L2.
  int tm, tn;
  {
    const int BAND = 8;
    int nb_full = tiles_m / BAND;
    int full_total = nb_full * BAND * tiles_n;
    if (wg < full_total) {
      int band = wg / (BAND * tiles_n);
      int r2 = wg % (BAND * tiles_n);
      tn = r2 / BAND;
      tm = band * BAND + r2 % BAND;
    } else {
      int r2 = wg - full_total;
      int bh = tiles_m - nb_full * BAND;
      tn = r2 / bh;
      tm = nb_full * BAND + r2 % bh;
    }
  }
  const long long brow = (long long)tm * GM_BM;
  const long long bcol = (long long)tn * GM_BN;

  const long long ldA = K * 2;  // bytes
  const char* gA = (const char*)A + brow * ldA;
  const char* gW = (const char*)W + bcol * ldA;

  // Per-lane swizzled stage offsets within a [256][GM_BK] tile: identical
  // for A and W (same chunk/lane pattern, same row stride). 4 chunks/wave.
  int soff[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int x = (wid * 4 + c) * 1024 + lane * 16;
    int p = swz(x);
    soff[c] = (p >> 7) * (int)ldA + (p & 127);
  }

  float4v acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int nkt = (int)(K / GM_BK);  // even (K % 128 == 0 checked by host)

  // ---- prologue: stage K-tiles 0 -> dbuf0 and 1 -> dbuf1 ----
  // Per K-tile per matrix: 32 chunks over 8 waves = 4 chunks/wave.
#pragma unroll
  for (int c = 0; c < 4; ++c) stage_chunk(gA, soff[c], sA[0], wid * 4 + c);
#pragma unroll
  for (int c = 0; c < 4; ++c) stage_chunk(gW, soff[c], sB[0], wid * 4 + c);
#pragma unroll
  for (int c = 0; c < 4; ++c)
    stage_chunk(gA + GM_BK * 2, soff[c], sA[1], wid * 4 + c);
#pragma unroll
  for (int c = 0; c < 4; ++c)
    stage_chunk(gW + GM_BK * 2, soff[c], sB[1], wid * 4 + c);
  // Gate dbuf0 (allow K-tile 1's 8 loads to stay in flight), then make the
  // landing visible to all waves.
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // Fragment read helpers. A frag for (m, ks): row = wr*128 + m*16 + lane%16,
  // 16 B at column byte (ks*32 + (lane/16)*8)*2. B frag for (n, ks):
  // "row" = wcol*64 + n*16 + lane%16 (W rows are output columns).
  const int fr_a = wr * 128 + (lane & 15);      // + m*16
  const int fr_b = wcol * 64 + (lane & 15);     // + n*16
  const int fkb = ((lane >> 4) * 8) * 2;        // + ks*64

// LDS byte addresses: addrspace(3) pointer values ARE byte offsets, so
// fragment reads can be inline-asm ds_read_b128 with integer addresses.
// Inline asm (not pointer loads) is REQUIRED for throughput here: with
// compiler-visible loads hipcc emits one s_waitcnt lgkmcnt(0) before the
// first MFMA of each phase, serializing the whole 4-12-read LDS burst
// with the 16-MFMA cluster (measured 43% MFMA util — LDS pipe and MFMA
// pipe are nearly balanced at ~2.3k vs ~2.5k cyc/CU/K-tile, so the
// serialization costs ~1.8x). With counted lgkmcnt(N) waits between
// MFMA sub-groups the read tail overlaps the MFMA head. Every wait is
// followed by sched_barrier(0): hipcc otherwise hoists register-only
// MFMAs past inline-asm waitcnts (guide rule 18).
  const unsigned lA0 = (unsigned)(unsigned long long)(L_AS char*)sA[0];
  const unsigned lB0 = (unsigned)(unsigned long long)(L_AS char*)sB[0];
  const unsigned aoff = (unsigned)(fr_a * 128 + fkb);
  const unsigned boff = (unsigned)(fr_b * 128 + fkb);

#define A_ADDR(d, m, ks) \
  (lA0 + (d) * (GM_BM * GM_BK * 2) + swz((int)aoff + (m) * 2048 + (ks) * 64))
#define B_ADDR(d, n, ks) \
  (lB0 + (d) * (GM_BM * GM_BK * 2) + swz((int)boff + (n) * 2048 + (ks) * 64))

#define DS_READ(dst, addr)                                                \
  if (!(ABL & 2)) {                                                       \
    asm volatile("ds_read_b128 %0, %1" : "=&v"(dst) : "v"(addr));         \
  }
#define WAIT_LGKM(n)                                                      \
  if (!(ABL & 2)) {                                                       \
    asm volatile("s_waitcnt lgkmcnt(" #n ")" ::: "memory");               \
    __builtin_amdgcn_sched_barrier(0);                                    \
  }
// rule-17 keepalive: when MFMA is ablated the fragment reads would be
// dead and DCE'd; this pins them live at zero cost.
#define KEEPALIVE(v0)                                                     \
  asm volatile("" ::"v"(v0))

  bf16x8 af[4][2];   // current mh group: 4 m-frags x 2 k-steps
  bf16x8 bn0[2][2];  // nh=0 col-frags (kept across the whole K-tile)
  bf16x8 bn1[2][2];  // nh=1 col-frags
  if (ABL & 2) {
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int k2 = 0; k2 < 2; ++k2) {
        const short8v z = {0, 0, 0, 0, 0, 0, 0, 0};
        af[i][k2] = (bf16x8)z;
        if (i < 2) { bn0[i][k2] = (bf16x8)z; bn1[i][k2] = (bf16x8)z; }
      }
  }

// 4 MFMAs: one m-fragment against both n-fragments of a quad, both k-steps
#define MFMA_MI(mh, nh, mi, breg)                                         \
  _Pragma("unroll") for (int ni = 0; ni < 2; ++ni) {                      \
    _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) {                    \
      if (ABL & 4) { KEEPALIVE(af[mi][ks]); KEEPALIVE(breg[ni][ks]); continue; } \
      acc[(mh)*4 + (mi)][(nh)*2 + ni] =                                   \
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(                        \
              af[mi][ks], breg[ni][ks], acc[(mh)*4 + (mi)][(nh)*2 + ni],  \
              0, 0, 0);                                                   \
    }                                                                     \
  }

// 8 MFMAs: one n-fragment against all four m-fragments, both k-steps
#define MFMA_NI(mh, nh, ni, breg)                                         \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi) {                      \
    _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) {                    \
      if (ABL & 4) { KEEPALIVE(af[mi][ks]); KEEPALIVE(breg[ni][ks]); continue; } \
      acc[(mh)*4 + mi][(nh)*2 + (ni)] =                                   \
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(                        \
              af[mi][ks], breg[ni][ks], acc[(mh)*4 + mi][(nh)*2 + (ni)],  \
              0, 0, 0);                                                   \
    }                                                                     \
  }

  const int niter = nkt >> 1;  // nkt is even
  for (int it = 0; it < niter; ++it) {
    const int pf0 = 2 * it + 2;  // K-tile staged into dbuf0 this iter
    const int pf1 = 2 * it + 3;  // K-tile staged into dbuf1 this iter
    const char* gA0 = gA + (long long)pf0 * (GM_BK * 2);
    const char* gW0 = gW + (long long)pf0 * (GM_BK * 2);
    const char* gA1 = gA + (long long)pf1 * (GM_BK * 2);
    const char* gW1 = gW + (long long)pf1 * (GM_BK * 2);

    // ---------- group d0: K-tile 2*it, phases 0..3 ----------
    // phase 0: quad (0,0); 12 reads in consumption order (af0, bn0,
    // af1..3); sub-group waits allow 6/4/2/0 reads outstanding.
    DS_READ(af[0][0], A_ADDR(0, 0, 0));
    DS_READ(af[0][1], A_ADDR(0, 0, 1));
    DS_READ(bn0[0][0], B_ADDR(0, 0, 0));
    DS_READ(bn0[0][1], B_ADDR(0, 0, 1));
    DS_READ(bn0[1][0], B_ADDR(0, 1, 0));
    DS_READ(bn0[1][1], B_ADDR(0, 1, 1));
    DS_READ(af[1][0], A_ADDR(0, 1, 0));
    DS_READ(af[1][1], A_ADDR(0, 1, 1));
    DS_READ(af[2][0], A_ADDR(0, 2, 0));
    DS_READ(af[2][1], A_ADDR(0, 2, 1));
    DS_READ(af[3][0], A_ADDR(0, 3, 0));
    DS_READ(af[3][1], A_ADDR(0, 3, 1));
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(6); MFMA_MI(0, 0, 0, bn0);
    WAIT_LGKM(4); MFMA_MI(0, 0, 1, bn0);
    WAIT_LGKM(2); MFMA_MI(0, 0, 2, bn0);
    WAIT_LGKM(0); MFMA_MI(0, 0, 3, bn0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 1: quad (0,1); 4 reads (bn1)
    DS_READ(bn1[0][0], B_ADDR(0, 2, 0));
    DS_READ(bn1[0][1], B_ADDR(0, 2, 1));
    DS_READ(bn1[1][0], B_ADDR(0, 3, 0));
    DS_READ(bn1[1][1], B_ADDR(0, 3, 1));
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(2); MFMA_NI(0, 1, 0, bn1);
    WAIT_LGKM(0); MFMA_NI(0, 1, 1, bn1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 2: quad (1,1); 8 A reads (mh1); stages B dbuf0 <- K-tile pf0
    // (B dbuf0's last ds_read was phase 1, complete before its end
    // barrier because phase 1 ends with WAIT_LGKM(0))
    DS_READ(af[0][0], A_ADDR(0, 4, 0));
    DS_READ(af[0][1], A_ADDR(0, 4, 1));
    DS_READ(af[1][0], A_ADDR(0, 5, 0));
    DS_READ(af[1][1], A_ADDR(0, 5, 1));
    DS_READ(af[2][0], A_ADDR(0, 6, 0));
    DS_READ(af[2][1], A_ADDR(0, 6, 1));
    DS_READ(af[3][0], A_ADDR(0, 7, 0));
    DS_READ(af[3][1], A_ADDR(0, 7, 1));
    if (!(ABL & 1) && pf0 < nkt) {
#pragma unroll
      for (int c = 0; c < 4; ++c) stage_chunk(gW0, soff[c], sB[0], wid * 4 + c);
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(6); MFMA_MI(1, 1, 0, bn1);
    WAIT_LGKM(4); MFMA_MI(1, 1, 1, bn1);
    WAIT_LGKM(2); MFMA_MI(1, 1, 2, bn1);
    WAIT_LGKM(0); MFMA_MI(1, 1, 3, bn1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 3: quad (1,0); no reads (af holds mh1, bn0 kept); stages A
    // dbuf0 <- pf0. End barrier gates dbuf1 (staged prev phases 6,7):
    // vmcnt(8) leaves this iter's phase-2/3 stage loads in flight.
    if (!(ABL & 1) && pf0 < nkt) {
#pragma unroll
      for (int c = 0; c < 4; ++c) stage_chunk(gA0, soff[c], sA[0], wid * 4 + c);
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    MFMA_MI(1, 0, 0, bn0);
    MFMA_MI(1, 0, 1, bn0);
    MFMA_MI(1, 0, 2, bn0);
    MFMA_MI(1, 0, 3, bn0);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // ---------- group d1: K-tile 2*it+1, phases 4..7 ----------
    DS_READ(af[0][0], A_ADDR(1, 0, 0));
    DS_READ(af[0][1], A_ADDR(1, 0, 1));
    DS_READ(bn0[0][0], B_ADDR(1, 0, 0));
    DS_READ(bn0[0][1], B_ADDR(1, 0, 1));
    DS_READ(bn0[1][0], B_ADDR(1, 1, 0));
    DS_READ(bn0[1][1], B_ADDR(1, 1, 1));
    DS_READ(af[1][0], A_ADDR(1, 1, 0));
    DS_READ(af[1][1], A_ADDR(1, 1, 1));
    DS_READ(af[2][0], A_ADDR(1, 2, 0));
    DS_READ(af[2][1], A_ADDR(1, 2, 1));
    DS_READ(af[3][0], A_ADDR(1, 3, 0));
    DS_READ(af[3][1], A_ADDR(1, 3, 1));
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(6); MFMA_MI(0, 0, 0, bn0);
    WAIT_LGKM(4); MFMA_MI(0, 0, 1, bn0);
    WAIT_LGKM(2); MFMA_MI(0, 0, 2, bn0);
    WAIT_LGKM(0); MFMA_MI(0, 0, 3, bn0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 5
    DS_READ(bn1[0][0], B_ADDR(1, 2, 0));
    DS_READ(bn1[0][1], B_ADDR(1, 2, 1));
    DS_READ(bn1[1][0], B_ADDR(1, 3, 0));
    DS_READ(bn1[1][1], B_ADDR(1, 3, 1));
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(2); MFMA_NI(0, 1, 0, bn1);
    WAIT_LGKM(0); MFMA_NI(0, 1, 1, bn1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 6: stages B dbuf1 <- pf1
    DS_READ(af[0][0], A_ADDR(1, 4, 0));
    DS_READ(af[0][1], A_ADDR(1, 4, 1));
    DS_READ(af[1][0], A_ADDR(1, 5, 0));
    DS_READ(af[1][1], A_ADDR(1, 5, 1));
    DS_READ(af[2][0], A_ADDR(1, 6, 0));
    DS_READ(af[2][1], A_ADDR(1, 6, 1));
    DS_READ(af[3][0], A_ADDR(1, 7, 0));
    DS_READ(af[3][1], A_ADDR(1, 7, 1));
    if (!(ABL & 1) && pf1 < nkt) {
#pragma unroll
      for (int c = 0; c < 4; ++c) stage_chunk(gW1, soff[c], sB[1], wid * 4 + c);
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    WAIT_LGKM(6); MFMA_MI(1, 1, 0, bn1);
    WAIT_LGKM(4); MFMA_MI(1, 1, 1, bn1);
    WAIT_LGKM(2); MFMA_MI(1, 1, 2, bn1);
    WAIT_LGKM(0); MFMA_MI(1, 1, 3, bn1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);

    // phase 7: stages A dbuf1 <- pf1; end barrier gates dbuf0
    if (!(ABL & 1) && pf1 < nkt) {
#pragma unroll
      for (int c = 0; c < 4; ++c) stage_chunk(gA1, soff[c], sA[1], wid * 4 + c);
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    MFMA_MI(1, 0, 0, bn0);
    MFMA_MI(1, 0, 1, bn0);
    MFMA_MI(1, 0, 2, bn0);
    MFMA_MI(1, 0, 3, bn0);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
  }

#undef A_ADDR
#undef B_ADDR
#undef DS_READ
#undef WAIT_LGKM
#undef MFMA_MI
#undef MFMA_NI

  // ---- epilogue: bias (+GELU) in fp32, bf16 convert, LDS bounce so the
  // global store is 16 B coalesced. Per-wave region: 128x64 bf16 = 16 KiB.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  unsigned short* sC = (unsigned short*)(smem + wid * 128 * 64 * 2);

  float bvals[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    if (HAS_BIAS) {
      bvals[n] =
          bf16_bits_to_f32(bias[bcol + wcol * 64 + n * 16 + (lane & 15)]);
    } else {
      bvals[n] = 0.0f;
    }
  }
#pragma unroll
  for (int m = 0; m < 8; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int row0 = m * 16 + (lane >> 4) * 4;
      const int col = n * 16 + (lane & 15);
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        float v = acc[m][n][e] + bvals[n];
        if (ACT == 1) v = gelu_erf(v);
        sC[(row0 + e) * 64 + col] = f32_to_bf16_bits(v);
      }
    }
  }
  __builtin_amdgcn_s_barrier();

  // read back + store: 16 iterations of 16 B per lane.
  const long long crow0 = brow + (long long)wr * 128;
  const long long ccol0 = bcol + (long long)wcol * 64;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    int off = j * 1024 + lane * 16;          // linear within the wave region
    int row = off >> 7;                      // 128 B per LDS row
    int colb = off & 127;
    uint4v v = *reinterpret_cast<const uint4v*>((const char*)sC + off);
    if (ABL & 8) {
      KEEPALIVE(v);
    } else {
      *reinterpret_cast<uint4v*>((char*)C + (crow0 + row) * N * 2 +
                                 ccol0 * 2 + colb) = v;
    }
  }
}


// ---------------------------------------------------------------------------
// Variant B: 96x256 tile, 4 waves, 3 workgroups/CU (the knn_mfma geometry).
//
// The 256x256 8-phase kernel above is a single-workgroup-per-CU design;
// hardware ablation (scripts/gemm_ablate.py) showed its costs are
// ADDITIVE — staging DMA, LDS fragment reads and the C store all
// serialize with the MFMA stream because nothing else is resident on
// the CU to overlap them. This variant trades per-tile efficiency for
// CO-RESIDENCY: 3 small workgroups per CU (44 KB LDS each, 12 waves)
// run phase-shifted, so one workgroup's staging and epilogue overlap
// another's MFMA cluster — the same occupancy lesson the kNN kernel
// established in round 1 (multi-WG occupancy IS the memory pipeline).
// ---------------------------------------------------------------------------

#define GB_BM 96
#define GB_BN 256
#define GB_BK 64
#define GB_NT 256
#define GB_MW (GB_BM / 16)

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GB_NT, 3) void k_gemm_nt96(
    const unsigned short* __restrict__ A,     // [M,K] bf16
    const unsigned short* __restrict__ W,     // [N,K] bf16
    const unsigned short* __restrict__ bias,  // [N] or nullptr
    unsigned short* __restrict__ C,           // [M,N] bf16
    int tiles_m, int tiles_n, long long K, long long N) {
  using namespace gemm_nt_detail;
  __shared__ __align__(16) char smem[GB_BM * GB_BK * 2 + GB_BN * GB_BK * 2];
  unsigned short* sA = (unsigned short*)smem;
  unsigned short* sB = (unsigned short*)(smem + GB_BM * GB_BK * 2);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wc = tid / WAVE;  // wave col 0..3 (64 output cols each)

  // XCD-bijective remap + band-major walk (same rationale as variant A:
  // a band of tm panels sweeps all tn so A stays L2/LLC resident).
  int nwg = tiles_m * tiles_n;
  int wg = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = wg & 7, slot = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  }
  int tm, tn;
  {
    const int BAND = 16;  // 16 x 96 rows = 3 MB A-band at K=1024
    int nb_full = tiles_m / BAND;
    int full_total = nb_full * BAND * tiles_n;
    if (wg < full_total) {
      int band = wg / (BAND * tiles_n);
      int r2 = wg % (BAND * tiles_n);
      tn = r2 / BAND;
      tm = band * BAND + r2 % BAND;
    } else {
      int r2 = wg - full_total;
      int bh = tiles_m - nb_full * BAND;
      tn = r2 / bh;
      tm = nb_full * BAND + r2 % bh;
    }
  }
  const long long brow = (long long)tm * GB_BM;
  const long long bcol = (long long)tn * GB_BN;
  const long long ld = K * 2;
  const char* gA = (const char*)A + brow * ld;
  const char* gW = (const char*)W + bcol * ld;

  float4v acc[GB_MW][4];
#pragma unroll
  for (int m = 0; m < GB_MW; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  for (long long kt = 0; kt < K; kt += GB_BK) {
    // stage A (96x64 = 12 KB: 12 chunks) + B (256x64 = 32 KB: 32 chunks)
    // with the pre-swizzled-source global_load_lds pattern (knn_mfma).
#pragma unroll
    for (int it = 0; it < 3; ++it) {
      int chunk = wc * 3 + it;
      int x = chunk * 1024 + lane * 16;
      int p = swz(x);
      const G_AS unsigned int* gp = (const G_AS unsigned int*)(
          gA + (long long)(p >> 7) * ld + kt * 2 + (p & 127));
      L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sA + chunk * 1024);
      __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
    }
#pragma unroll
    for (int it = 0; it < 8; ++it) {
      int chunk = wc * 8 + it;
      int x = chunk * 1024 + lane * 16;
      int p = swz(x);
      const G_AS unsigned int* gp = (const G_AS unsigned int*)(
          gW + (long long)(p >> 7) * ld + kt * 2 + (p & 127));
      L_AS unsigned int* lp = (L_AS unsigned int*)((char*)sB + chunk * 1024);
      __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
    }
    __syncthreads();

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
      for (int m = 0; m < GB_MW; ++m) {
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

  // epilogue: bias (+GELU), bf16, LDS bounce in two 48-row half-passes
  // ([48][256] bf16 = 24 KB, fits the freed staging space) for 16 B
  // coalesced global stores.
  float bvals[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    bvals[n] = HAS_BIAS
        ? bf16_bits_to_f32(bias[bcol + wc * 64 + n * 16 + (lane & 15)])
        : 0.0f;
  }
  unsigned short* sC = (unsigned short*)smem;  // [48][256]
#pragma unroll
  for (int half = 0; half < 2; ++half) {
#pragma unroll
    for (int m = 0; m < 3; ++m) {
      const int mi = half * 3 + m;
      const int row0 = m * 16 + (lane >> 4) * 4;
      const int col = wc * 64 + (lane & 15);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          float v = acc[mi][n][e] + bvals[n];
          if (ACT == 1) v = gelu_erf(v);
          sC[(row0 + e) * 256 + col + n * 16] = f32_to_bf16_bits(v);
        }
      }
    }
    __syncthreads();
    // store 48x256 bf16 = 24 KB: each thread 6 x 16 B
    const long long grow0 = brow + half * 48;
#pragma unroll
    for (int j = 0; j < 6; ++j) {
      int off = j * 4096 + tid * 16;          // 512 B per LDS row
      int row = off >> 9;
      int colb = off & 511;
      uint4v v = *reinterpret_cast<const uint4v*>((const char*)sC + off);
      *reinterpret_cast<uint4v*>((char*)C + (grow0 + row) * N * 2 +
                                 bcol * 2 + colb) = v;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// host wrapper
// ---------------------------------------------------------------------------
at::Tensor gemm_nt(at::Tensor a, at::Tensor w, c10::optional<at::Tensor> bias,
                   long long act) {
  TORCH_CHECK(a.is_cuda() && a.dim() == 2 && a.is_contiguous() &&
                  a.scalar_type() == at::kBFloat16,
              "gemm_nt: A must be contiguous 2D bf16 CUDA");
  TORCH_CHECK(w.is_cuda() && w.dim() == 2 && w.is_contiguous() &&
                  w.scalar_type() == at::kBFloat16,
              "gemm_nt: W must be contiguous 2D bf16 CUDA");
  long long M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "gemm_nt: K mismatch");
  // kernel choice: variant B (96x256, 3 WGs/CU) by default — co-residency
  // beats the 1-WG 256x256 template on the encoder shapes (ablation in
  // scripts/gemm_ablate.py); NORNICDB_GEMM_KERNEL=256 forces variant A.
  const char* kv = getenv("NORNICDB_GEMM_KERNEL");
  bool use96 = !(kv && atoi(kv) == 256) && (M % GB_BM == 0);
  TORCH_CHECK(use96 ? (M % GB_BM == 0) : (M % GM_BM == 0),
              "gemm_nt: M tiling (python pads)");
  TORCH_CHECK(N % GM_BN == 0, "gemm_nt: N % 256 == 0 (python pads)");
  TORCH_CHECK(K % (use96 ? GB_BK : 2 * GM_BK) == 0, "gemm_nt: K tiling");
  const unsigned short* bptr = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0) {
    TORCH_CHECK(bias->is_cuda() && bias->is_contiguous() &&
                    bias->scalar_type() == at::kBFloat16 && bias->numel() == N,
                "gemm_nt: bias must be contiguous bf16 [N]");
    bptr = (const unsigned short*)bias->data_ptr();
  }
  TORCH_CHECK(act <= 1, "gemm_nt: act in {0: none, 1: gelu, <0: ablation probe}");

  auto c = at::empty({M, N}, a.options());
  auto stream = at::hip::getCurrentHIPStream().stream();

  const unsigned short* ap = (const unsigned short*)a.data_ptr();
  const unsigned short* wp = (const unsigned short*)w.data_ptr();
  unsigned short* cp = (unsigned short*)c.data_ptr();

  if (use96 && act >= 0) {
    int tm96 = (int)(M / GB_BM), tn96 = (int)(N / GB_BN);
    dim3 g96(tm96 * tn96);
    auto launch96 = [&](auto kern) {
      hipLaunchKernelGGL(kern, g96, dim3(GB_NT), 0, stream, ap, wp, bptr, cp,
                         tm96, tn96, K, N);
    };
    if (act == 1) {
      if (bptr) launch96(k_gemm_nt96<1, true>);
      else launch96(k_gemm_nt96<1, false>);
    } else {
      if (bptr) launch96(k_gemm_nt96<0, true>);
      else launch96(k_gemm_nt96<0, false>);
    }
    HIP_CHECK_LAST();
    return c;
  }
  int tiles_m = (int)(M / GM_BM), tiles_n = (int)(N / GM_BN);
  dim3 grid(tiles_m * tiles_n);

  if (act < 0) {
    // ablation probe (timing only; output undefined for abl != 0):
    // act = -(1 + abl_bits)
    int abl = -(int)act - 1;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(GM_NTHREADS), 0, stream, ap, wp,
                         (const unsigned short*)nullptr, cp, tiles_m,
                         tiles_n, K, 0, N);
    };
    switch (abl) {
      case 1: launch(k_gemm_nt<0, false, 1>); break;
      case 2: launch(k_gemm_nt<0, false, 2>); break;
      case 3: launch(k_gemm_nt<0, false, 3>); break;
      case 4: launch(k_gemm_nt<0, false, 4>); break;
      case 6: launch(k_gemm_nt<0, false, 6>); break;
      case 8: launch(k_gemm_nt<0, false, 8>); break;
      case 12: launch(k_gemm_nt<0, false, 12>); break;
      default: launch(k_gemm_nt<0, false, 0>); break;
    }
    HIP_CHECK_LAST();
    return c;
  }
  if (act == 1) {
    if (bptr)
      hipLaunchKernelGGL((k_gemm_nt<1, true>), grid, dim3(GM_NTHREADS), 0,
                         stream, ap, wp, bptr, cp, tiles_m, tiles_n, K, 0, N);
    else
      hipLaunchKernelGGL((k_gemm_nt<1, false>), grid, dim3(GM_NTHREADS), 0,
                         stream, ap, wp, bptr, cp, tiles_m, tiles_n, K, 0, N);
  } else {
    if (bptr)
      hipLaunchKernelGGL((k_gemm_nt<0, true>), grid, dim3(GM_NTHREADS), 0,
                         stream, ap, wp, bptr, cp, tiles_m, tiles_n, K, 0, N);
    else
      hipLaunchKernelGGL((k_gemm_nt<0, false>), grid, dim3(GM_NTHREADS), 0,
                         stream, ap, wp, bptr, cp, tiles_m, tiles_n, K, 0, N);
  }
  HIP_CHECK_LAST();
  return c;
}
