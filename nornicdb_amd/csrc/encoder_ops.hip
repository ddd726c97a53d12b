// Fused encoder ops for the bge-m3 (XLM-R) forward pass.
//
// Replaces the reference's llama.cpp/ggml encoder internals
// (reference pkg/localllm/llama.go:104-180) with hand-written CDNA4
// kernels:
//   - k_add_layernorm      : y = LN(a + b) * gamma + beta   (bf16, fp32 stats)
//   - k_bias_gelu          : y = gelu_erf(x + bias)          (bf16)
//   - k_mean_pool_l2norm   : masked mean over S + L2 norm    (bf16 -> fp32)
//   - k_flash_attn_nc      : non-causal flash attention, head_dim 64,
//                            16x16x32 MFMA QK^T and PV, online softmax,
//                            V transposed at stage time in LDS.
// All memory-bound kernels use short8 (16 B/lane) vector loads per the
// CDNA4 guide (G13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define G_AS __attribute__((address_space(1)))
#define L_AS __attribute__((address_space(3)))

// ---------------------------------------------------------------------------
// y = LayerNorm(a + b) * gamma + beta. Rows = tokens, D % 8 == 0, D <= 8192.
// One wave per row.
// ---------------------------------------------------------------------------
__global__ void k_add_layernorm(const unsigned short* __restrict__ a,
                                const unsigned short* __restrict__ b,
                                const unsigned short* __restrict__ gamma,
                                const unsigned short* __restrict__ beta,
                                unsigned short* __restrict__ y,
                                long long rows, int d, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int wpb = blockDim.x / WAVE;
  const long long w0 = (long long)blockIdx.x * wpb + wid;
  const long long tw = (long long)gridDim.x * wpb;

  for (long long row = w0; row < rows; row += tw) {
    const unsigned short* pa = a + row * d;
    const unsigned short* pb = b ? b + row * d : nullptr;
    float sum = 0.f, sq = 0.f;
    // cache the summed row in registers: up to 8192/64 = 128 floats... too
    // many; re-read instead (L2-hot).
    for (int c = lane * 8; c < d; c += WAVE * 8) {
      short8v va = *reinterpret_cast<const short8v*>(pa + c);
      short8v vb;
      if (pb) vb = *reinterpret_cast<const short8v*>(pb + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_f32((unsigned short)va[j]);
        if (pb) f += bf16_bits_to_f32((unsigned short)vb[j]);
        sum += f;
        sq += f * f;
      }
    }
    sum = wave_reduce_sum(sum);
    sq = wave_reduce_sum(sq);
    float mean = sum / d;
    float inv = rsqrtf(fmaxf(sq / d - mean * mean, 0.f) + eps);
    unsigned short* py = y + row * d;
    for (int c = lane * 8; c < d; c += WAVE * 8) {
      short8v va = *reinterpret_cast<const short8v*>(pa + c);
      short8v vb;
      if (pb) vb = *reinterpret_cast<const short8v*>(pb + c);
      short8v vg = *reinterpret_cast<const short8v*>(gamma + c);
      short8v vbe = *reinterpret_cast<const short8v*>(beta + c);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_f32((unsigned short)va[j]);
        if (pb) f += bf16_bits_to_f32((unsigned short)vb[j]);
        float g = bf16_bits_to_f32((unsigned short)vg[j]);
        float be = bf16_bits_to_f32((unsigned short)vbe[j]);
        o[j] = (short)f32_to_bf16_bits((f - mean) * inv * g + be);
      }
      *reinterpret_cast<short8v*>(py + c) = o;
    }
  }
}

// ---------------------------------------------------------------------------
// y = gelu(x + bias), exact erf form (XLM-R GELU). x [rows][d], bias [d].
// Grid-stride elementwise, short8 vectorized.
// ---------------------------------------------------------------------------
__global__ void k_bias_gelu(const unsigned short* __restrict__ x,
                            const unsigned short* __restrict__ bias,
                            unsigned short* __restrict__ y,
                            long long total, int d) {
  long long i0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long i = i0; i < total; i += stride) {
    short8v v = *reinterpret_cast<const short8v*>(x + i);
    int c = (int)(i % d);
    short8v vb = *reinterpret_cast<const short8v*>(bias + c);
    short8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_bits_to_f32((unsigned short)v[j]) +
                bf16_bits_to_f32((unsigned short)vb[j]);
      float g = 0.5f * f * (1.0f + erff(f * 0.70710678f));
      o[j] = (short)f32_to_bf16_bits(g);
    }
    *reinterpret_cast<short8v*>(y + i) = o;
  }
}

// ---------------------------------------------------------------------------
// Masked mean-pool over sequence + L2 normalize: x [B][S][D] bf16,
// mask [B][S] (int32, 1=real), out [B][D] fp32 unit-norm.
// One block per batch row; lanes split D.
// ---------------------------------------------------------------------------
__global__ void k_mean_pool_l2norm(const unsigned short* __restrict__ x,
                                   const int* __restrict__ mask,
                                   float* __restrict__ out,
                                   int bsz, int s, int d) {
  const int b = blockIdx.x;
  if (b >= bsz) return;
  extern __shared__ float s_acc[];  // [d]
  for (int c = threadIdx.x; c < d; c += blockDim.x) s_acc[c] = 0.f;
  __syncthreads();
  int count = 0;
  for (int t = 0; t < s; ++t) {
    if (mask && mask[b * s + t] == 0) continue;
    count++;
    const unsigned short* p = x + ((long long)b * s + t) * d;
    for (int c = threadIdx.x * 8; c < d; c += blockDim.x * 8) {
      short8v v = *reinterpret_cast<const short8v*>(p + c);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s_acc[c + j] += bf16_bits_to_f32((unsigned short)v[j]);
    }
    __syncthreads();
  }
  // mean + norm
  float inv_n = 1.0f / max(count, 1);
  float ss = 0.f;
  for (int c = threadIdx.x; c < d; c += blockDim.x) {
    float m = s_acc[c] * inv_n;
    s_acc[c] = m;
    ss += m * m;
  }
  __shared__ float s_red[8];
  ss = block_reduce_sum(ss, s_red);
  float inv = rsqrtf(ss + 1e-12f);
  for (int c = threadIdx.x; c < d; c += blockDim.x)
    out[(long long)b * d + c] = s_acc[c] * inv;
}

// ---------------------------------------------------------------------------
// Non-causal flash attention forward, head_dim 64, bf16, no mask
// (full attention; padded batches fall back to torch sdpa in python).
//
// q,k,v: [B, H, S, 64] bf16 contiguous. out: same.
// Block: 256 thr = 4 waves; one (b, h, 64-query tile) per block.
// Wave w owns q rows [w*16, w*16+16). K-tile loop of 64 keys:
//   K staged linear (global_load_lds, tile is contiguous 8 KB),
//   V staged TRANSPOSED via registers (VT[d][k], 2-way-bank-free),
//   QK^T: 2x mfma 16x16x32 per 16-key subtile (B-frag direct from K rows),
//   online softmax in C-fragment registers (16-lane group reduces),
//   P -> LDS (bf16) -> A-frags; PV: mfma with B-frags from VT.
// ---------------------------------------------------------------------------
#define FA_D 64
#define FA_KT 64
#define FA_QT 64
#define FA_PSTRIDE 64   // measured: 64 beats 72 by 15% (208 vs 181 TF; scripts/attn_pad.hip sweep)
#define FA_VSTRIDE 80   // measured best with gather-V staging (232 vs 226 TF at 72)

__global__ __launch_bounds__(256, 2) void k_flash_attn_nc(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    int n_heads, int s, float scale,
    long long q_bs, long long q_ss,   // Q batch/seq strides (elements)
    long long k_bs, long long k_ss,
    long long v_bs, long long v_ss) {
  // Layout: Q/K/V are [B, S, H, 64] views with arbitrary batch/seq strides
  // (head stride == 64, last dim contiguous) so the qkv projection output
  // feeds in with ZERO transposes; O is contiguous [B, S, H, 64].
  __shared__ unsigned short sK[FA_KT * FA_D];          // 8 KB, linear
  __shared__ unsigned short sVT[FA_D * FA_VSTRIDE];    // 9 KB, transposed
  __shared__ unsigned short sP[4][16 * FA_PSTRIDE];    // 4 x 2.25 KB

  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int b = blockIdx.x / n_heads;
  const int h = blockIdx.x % n_heads;
  const int q0 = blockIdx.y * FA_QT;
  const long long hoff = (long long)h * FA_D;

  const int lq = lane & 15;           // row-in-16 for A-frags / col for C
  const int lk8 = (lane >> 4) * 8;    // k-offset for A/B frags

  // ---- load Q fragments for this wave's 16 rows (held for whole kernel) ----
  // A-frag for QK^T: lane holds Q[q=lq][d = lk8 + ks*32 .. +8], ks = 0,1
  bf16x8 qf[2];
  {
    const unsigned short* qp =
        Q + b * q_bs + (long long)(q0 + wid * 16 + lq) * q_ss + hoff;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      qf[ks] = (bf16x8)(*reinterpret_cast<const short8v*>(qp + ks * 32 + lk8));
  }

  // online softmax state: per lane, 4 q-rows (r = 0..3 of its 16-lane group)
  float m_run[4], l_run[4];
  float4v acc[4];  // output [16q][64d]: 4 d-subtiles x 4 regs
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int nn = 0; nn < 4; ++nn) acc[nn] = {0.f, 0.f, 0.f, 0.f};

  const float log2e = 1.44269504f;

  for (int k0 = 0; k0 < s; k0 += FA_KT) {
    // ---- stage K tile: 64 rows x 128 B (strided source, linear LDS) ----
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
    // ---- stage V transposed: gather 8 consecutive k at fixed d, ONE
    // b128 LDS write per octet. The strided global reads hit L1/L2 (the
    // V tile is 8 KB); swapping the cost off the LDS write port was
    // +5% within-probe on top of the stride fix (scripts/attn_pad.hip:
    // 216 -> 227 TF, refchecked).
    {
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
        *reinterpret_cast<short8v*>(sVT + dd * FA_VSTRIDE + kk) = o;
      }
    }
    __syncthreads();

    // ---- QK^T: 4 k-subtiles of 16 ----
    float4v sfrag[4];
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {
      float4v c4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        // B-frag: lane holds K[k = nn*16 + lq][d = lk8 + ks*32 .. +8]
        bf16x8 bf = (bf16x8)(*reinterpret_cast<const short8v*>(
            sK + (nn * 16 + lq) * FA_D + ks * 32 + lk8));
        c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], bf, c4, 0, 0, 0);
      }
      sfrag[nn] = c4;
    }

    // ---- online softmax ----
    // sfrag[nn][r] = S[q = (lane>>4)*4 + r][k = nn*16 + lq] * (pending scale)
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -1e30f;
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) mx = fmaxf(mx, sfrag[nn][r]);
      // reduce across the 16 lanes of this group
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
    // p = exp(s*scale - m); accumulate l; write P to LDS bf16
    float lsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f((sfrag[nn][r] * scale - m_run[r]) * log2e);
        lsum[r] += p;
        sP[wid][((lane >> 4) * 4 + r) * FA_PSTRIDE + nn * 16 + lq] =
            f32_to_bf16_bits(p);
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
    // rescale existing output acc
#pragma unroll
    for (int nn = 0; nn < 4; ++nn)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[nn][r] *= rescale[r];

    __syncthreads();  // P visible to own wave only, but VT/K rewrite below

    // ---- PV: A = P [16q x 64k] from sP, B = V^T from sVT ----
#pragma unroll
    for (int nn = 0; nn < 4; ++nn) {       // d-subtile
      float4v c4 = acc[nn];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {     // k-dim halves of 32
        bf16x8 af = (bf16x8)(*reinterpret_cast<const short8v*>(
            sP[wid] + lq * FA_PSTRIDE + ks * 32 + lk8));
        bf16x8 bf = (bf16x8)(*reinterpret_cast<const short8v*>(
            sVT + (nn * 16 + lq) * FA_VSTRIDE + ks * 32 + lk8));
        c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c4, 0, 0, 0);
      }
      acc[nn] = c4;
    }
    __syncthreads();
  }

  // ---- epilogue: normalize by l, write O [B,S,H,64] contiguous ----
  const long long o_ss = (long long)n_heads * FA_D;
  const long long obase =
      ((long long)b * s + q0 + wid * 16) * o_ss + hoff;
#pragma unroll
  for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      float v = acc[nn][r] / fmaxf(l_run[r], 1e-20f);
      O[obase + row * o_ss + nn * 16 + lq] = f32_to_bf16_bits(v);
    }
  }
}

// ===========================================================================
// Host wrappers
// ===========================================================================

static inline hipStream_t enc_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

at::Tensor add_layernorm(at::Tensor a, c10::optional<at::Tensor> b,
                         at::Tensor gamma, at::Tensor beta, double eps) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  auto a2 = a.contiguous();
  long long rows = a2.numel() / a2.size(-1);
  int d = (int)a2.size(-1);
  TORCH_CHECK(d % 8 == 0 && d <= 8192, "add_layernorm: D % 8 == 0, D <= 8192");
  at::Tensor y = at::empty_like(a2);
  const unsigned short* bp = nullptr;
  at::Tensor bc;
  if (b.has_value()) {
    bc = b->contiguous();
    TORCH_CHECK(bc.sizes() == a2.sizes());
    bp = (const unsigned short*)bc.data_ptr();
  }
  auto g = gamma.contiguous().to(at::kBFloat16);
  auto be = beta.contiguous().to(at::kBFloat16);
  int blocks = (int)std::min<long long>((rows + 3) / 4, 4096);
  hipLaunchKernelGGL(k_add_layernorm, dim3(blocks), dim3(256), 0, enc_stream(),
                     (const unsigned short*)a2.data_ptr(), bp,
                     (const unsigned short*)g.data_ptr(),
                     (const unsigned short*)be.data_ptr(),
                     (unsigned short*)y.data_ptr(), rows, d, (float)eps);
  HIP_CHECK_LAST();
  return y.view(a.sizes());
}

at::Tensor bias_gelu(at::Tensor x, at::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  auto x2 = x.contiguous();
  int d = (int)x2.size(-1);
  TORCH_CHECK(d % 8 == 0, "bias_gelu: D % 8 == 0");
  auto b = bias.contiguous().to(at::kBFloat16);
  at::Tensor y = at::empty_like(x2);
  long long total = x2.numel();
  int blocks = (int)std::min<long long>((total / 8 + 255) / 256, 4096);
  hipLaunchKernelGGL(k_bias_gelu, dim3(blocks), dim3(256), 0, enc_stream(),
                     (const unsigned short*)x2.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     (unsigned short*)y.data_ptr(), total, d);
  HIP_CHECK_LAST();
  return y.view(x.sizes());
}

at::Tensor mean_pool_l2norm(at::Tensor x, c10::optional<at::Tensor> mask) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.scalar_type() == at::kBFloat16);
  auto x2 = x.contiguous();
  int bsz = (int)x2.size(0), s = (int)x2.size(1), d = (int)x2.size(2);
  TORCH_CHECK(d % 8 == 0);
  const int* mp = nullptr;
  at::Tensor mc;
  if (mask.has_value()) {
    mc = mask->to(at::kInt).contiguous();
    TORCH_CHECK(mc.size(0) == bsz && mc.size(1) == s);
    mp = mc.data_ptr<int>();
  }
  at::Tensor out = at::empty({bsz, d}, x2.options().dtype(at::kFloat));
  size_t lds = (size_t)d * sizeof(float);
  hipLaunchKernelGGL(k_mean_pool_l2norm, dim3(bsz), dim3(256), lds,
                     enc_stream(), (const unsigned short*)x2.data_ptr(), mp,
                     out.data_ptr<float>(), bsz, s, d);
  HIP_CHECK_LAST();
  return out;
}

at::Tensor flash_attn_nc(at::Tensor q, at::Tensor k, at::Tensor v) {
  // q,k,v: [B, S, H, 64] bf16 VIEWS — head stride must be 64 and the last
  // dim contiguous; batch/seq strides are free (so qkv.unbind(2) feeds in
  // without any transpose/copy). Returns contiguous [B, S, H, 64].
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16,
              "flash_attn_nc: q [B,S,H,D] bf16");
  int B = (int)q.size(0), S = (int)q.size(1), H = (int)q.size(2),
      D = (int)q.size(3);
  TORCH_CHECK(D == 64, "flash_attn_nc supports head_dim 64");
  TORCH_CHECK(S % 64 == 0, "flash_attn_nc needs S % 64 == 0");
  for (auto* t : {&q, &k, &v}) {
    TORCH_CHECK(t->stride(3) == 1 && t->stride(2) == D,
                "flash_attn_nc: head stride must be D, last dim contiguous");
  }
  at::Tensor o = at::empty({B, S, H, D}, q.options());
  float scale = 1.0f / sqrtf((float)D);
  dim3 grid(B * H, S / 64);
  hipLaunchKernelGGL(k_flash_attn_nc, grid, dim3(256), 0, enc_stream(),
                     (const unsigned short*)q.data_ptr(),
                     (const unsigned short*)k.data_ptr(),
                     (const unsigned short*)v.data_ptr(),
                     (unsigned short*)o.data_ptr(), H, S, scale,
                     q.stride(0), q.stride(1), k.stride(0), k.stride(1),
                     v.stride(0), v.stride(1));
  HIP_CHECK_LAST();
  return o;
}
