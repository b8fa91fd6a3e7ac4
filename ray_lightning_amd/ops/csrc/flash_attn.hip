// Flash attention (causal, hs=64, bf16) — hand-written CDNA4 MFMA
// kernels. AOTriton's SDPA kernels on gfx950 measure ~186 TF/s
// causal-effective forward and ~115 TF/s backward on the GPT-2 shapes
// (profiles/r01_gpt2xl_1gpu_baseline.md) — far below the 2.5 PF bf16
// MFMA ceiling; these kernels target a multiple of that with the
// guide's plain-HIP GEMM idioms (LDS-staged tiles, 16x16x32 MFMA,
// flash online softmax).
//
// Layout: q, k, v are [B, H, T, 64] bf16 contiguous. Causal only.
// Fragment maps assumed here are verified by mfma_probe
// (tests/test_flash_attn_gpu.py) — cdna4_isa.md is not available in
// this environment, so the maps follow the CDNA3 pattern at 2xK:
//   A[16Mx32K]: lane l elem i -> row l&15,  k (l>>4)*8+i
//   B[32Kx16N]: lane l elem i -> col l&15,  k (l>>4)*8+i
//   C  [16x16]: lane l reg  r -> row (l>>4)*4+r, col l&15
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int HS = 64;      // head size (fixed)
constexpr int BM = 64;      // query rows per workgroup (16 per wave)
constexpr int BN = 64;      // key/value rows per tile
constexpr float kNegInf = -1e30f;

// rowwise reduce across the 16 lanes that share a C-row (lanes of one
// 16-lane group hold cols 0..15 of a tile).
__device__ __forceinline__ float row_reduce_max(float v) {
  #pragma unroll
  for (int w = 8; w >= 1; w >>= 1)
    v = fmaxf(v, __shfl_xor(v, w, 64));
  return v;
}
__device__ __forceinline__ float row_reduce_sum(float v) {
  #pragma unroll
  for (int w = 8; w >= 1; w >>= 1)
    v += __shfl_xor(v, w, 64);
  return v;
}

// q/k fragment loader: 16 bytes per lane from a [rows, HS] row-major
// bf16 tile. row = base_row + (l&15), k-chunk = kk*32 + (l>>4)*8.
__device__ __forceinline__ bf16x8 load_frag_rowmajor(
    const __hip_bfloat16* __restrict__ p, int base_row, int kk,
    int lane, long row_stride) {
  const long row = base_row + (lane & 15);
  const long off = row * row_stride + kk * 32 + (lane >> 4) * 8;
  bf16x8 v;
  *reinterpret_cast<int4*>(&v) =
      *reinterpret_cast<const int4*>(p + off);
  return v;
}

// ---------------------------------------------------------------------
// forward: one workgroup = BM query rows of one (b, h); 4 waves, each
// owns 16 query rows. K-tiles stream; V staged transposed in LDS; P
// staged per-wave in LDS between the softmax and the PV MFMAs.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void flash_fwd_kernel(
    const __hip_bfloat16* __restrict__ Q,
    const __hip_bfloat16* __restrict__ K,
    const __hip_bfloat16* __restrict__ V,
    __hip_bfloat16* __restrict__ O, float* __restrict__ LSE,
    int T, float scale) {
  const int bh = blockIdx.y;
  const int qtile = blockIdx.x;
  const int qm0 = qtile * BM;
  if (qm0 >= T) return;
  const long base = (long)bh * T * HS;
  const __hip_bfloat16* q = Q + base;
  const __hip_bfloat16* k = K + base;
  const __hip_bfloat16* v = V + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow0 = qm0 + wave * 16;   // this wave's first query row

  // LDS: V^T image [HS][BN] bf16 (8 KB) + per-wave P [16][BN] (2 KB x4)
  __shared__ __bf16 lds_vt[HS * BN];
  __shared__ __bf16 lds_p[4][16 * BN];

  // Q fragments for this wave's 16 rows (2 k-chunks), kept in regs.
  bf16x8 qf[2];
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk)
    qf[kk] = load_frag_rowmajor(q, wrow0, kk, lane, HS);

  // online-softmax state: each lane carries its 4 C-rows' stats
  float m_i[4], l_i[4];
  #pragma unroll
  for (int r = 0; r < 4; ++r) { m_i[r] = kNegInf; l_i[r] = 0.f; }
  // O accumulators: 4 hs-tiles x f32x4
  f32x4 o_acc[4];
  #pragma unroll
  for (int n = 0; n < 4; ++n) o_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const float l2e = 1.4426950408889634f * scale;  // exp2 domain
  const int kv_end = qm0 + BM;  // causal: tiles fully above the diag
  for (int kn0 = 0; kn0 < kv_end && kn0 < T; kn0 += BN) {
    // --- stage V^T into LDS (cooperative, coalesced read) ---------
    // thread t reads V[r][c] with r = t/8, c8 = (t%8)*8 (8 cols), and
    // writes transposed.
    {
      const int r = threadIdx.x >> 3;          // 0..31 x2 iterations
      const int c0 = (threadIdx.x & 7) * 8;
      #pragma unroll
      for (int rep = 0; rep < 2; ++rep) {
        const int row = r + rep * 32;
        bf16x8 vv;
        *reinterpret_cast<int4*>(&vv) = *reinterpret_cast<const int4*>(
            v + (long)(kn0 + row) * HS + c0);
        #pragma unroll
        for (int i = 0; i < 8; ++i)
          lds_vt[(c0 + i) * BN + row] = vv[i];
      }
    }
    __syncthreads();

    // --- S = Q K^T for this wave's 16 rows, 4 key tiles ------------
    f32x4 s_acc[4];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      s_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        // B fragment: key row = kn0 + 16n + (l&15); contiguous hs
        bf16x8 bf = load_frag_rowmajor(k, kn0 + 16 * n, kk, lane, HS);
        s_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qf[kk], bf, s_acc[n], 0, 0, 0);
      }
    }

    // --- causal mask (diagonal tile only) ---------------------------
    const int col0 = lane & 15;
    if (kn0 + BN > wrow0) {  // some keys may exceed some query rows
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow = wrow0 + (lane >> 4) * 4 + r;
          const int key = kn0 + 16 * n + col0;
          if (key > qrow) s_acc[n][r] = kNegInf;
        }
      }
    }

    // --- online softmax --------------------------------------------
    float p_scale[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(s_acc[0][r], s_acc[1][r]),
                       fmaxf(s_acc[2][r], s_acc[3][r]));
      mx = row_reduce_max(mx);
      const float m_new = fmaxf(m_i[r], mx);
      const float corr = exp2f((m_i[r] - m_new) * l2e);
      float rowsum = 0.f;
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        const float p = (s_acc[n][r] <= kNegInf * 0.5f)
            ? 0.f : exp2f((s_acc[n][r] - m_new) * l2e);
        s_acc[n][r] = p;  // reuse as P
        rowsum += p;
      }
      rowsum = row_reduce_sum(rowsum);
      l_i[r] = l_i[r] * corr + rowsum;
      m_i[r] = m_new;
      p_scale[r] = corr;
    }
    // rescale O accumulators by corr (per C-row r)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        o_acc[n][r] *= p_scale[r];

    // --- stage P to LDS in A-fragment layout ------------------------
    // P element (row=(l>>4)*4+r, col=16n+(l&15)) -> lds_p[w][row][col]
    __bf16* pw = lds_p[wave];
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[((lane >> 4) * 4 + r) * BN + 16 * n + col0] =
            (__bf16)s_acc[n][r];
    // wave-local staging: P producers == P consumers (same wave), so a
    // wave-level LDS visibility is enough; s_waitcnt lgkmcnt is implied
    // by the reads below on the same wave.
    __builtin_amdgcn_s_waitcnt(0);  // drain LDS writes for this wave

    // --- O += P · V  (contract over BN: 2 k-chunks) -----------------
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      // A fragment of P: row l&15, k (l>>4)*8+i (row-major [16][BN])
      bf16x8 pa;
      *reinterpret_cast<int4*>(&pa) = *reinterpret_cast<const int4*>(
          pw + (lane & 15) * BN + kk * 32 + (lane >> 4) * 8);
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        // B fragment of V: col = hs = 16n+(l&15), k = key;
        // from V^T image: row (16n + l&15) of lds_vt, contiguous keys
        bf16x8 vb;
        *reinterpret_cast<int4*>(&vb) = *reinterpret_cast<const int4*>(
            lds_vt + (16 * n + (lane & 15)) * BN + kk * 32 +
            (lane >> 4) * 8);
        o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, vb, o_acc[n], 0, 0, 0);
      }
    }
    __syncthreads();  // protect lds_vt before next tile overwrites
  }

  // --- epilogue: O / l, store bf16 + LSE ----------------------------
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = wrow0 + (lane >> 4) * 4 + r;
    if (qrow >= T) continue;
    const float inv_l = (l_i[r] > 0.f) ? 1.f / l_i[r] : 0.f;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      O[base + (long)qrow * HS + 16 * n + (lane & 15)] =
          __float2bfloat16(o_acc[n][r] * inv_l);
    if ((lane & 15) == 0)
      LSE[(long)bh * T + qrow] =
          m_i[r] * scale + logf(fmaxf(l_i[r], 1e-30f));
  }
}

}  // namespace

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q,
                                          torch::Tensor k,
                                          torch::Tensor v,
                                          double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 &&
              q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous(), "flash_attn_fwd: bf16 contiguous");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == HS,
              "flash_attn_fwd: [B,H,T,64] expected");
  const int B = (int)q.size(0), H = (int)q.size(1), T = (int)q.size(2);
  TORCH_CHECK(T % BM == 0, "flash_attn_fwd: T % 64 == 0");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, T},
                          q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid(T / BM, B * H);
  hipLaunchKernelGGL(flash_fwd_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(o.data_ptr()),
                     lse.data_ptr<float>(), T, (float)scale);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "flash_fwd: ", hipGetErrorString(e));
  return {o, lse};
}

// =====================================================================
// backward
// =====================================================================
namespace {

// D = rowsum(dO * O), fp32 [BH, T]
__global__ __launch_bounds__(256) void flash_bwd_pre_kernel(
    const __hip_bfloat16* __restrict__ dO,
    const __hip_bfloat16* __restrict__ O, float* __restrict__ D,
    long total_rows) {
  // one wave per 16 rows: lane handles row = base + lane/4, cols
  // (lane%4)*16 .. +16
  const long row = (long)blockIdx.x * 64 + (threadIdx.x >> 2);
  if (row >= total_rows) return;
  const int c0 = (threadIdx.x & 3) * 16;
  float s = 0.f;
  #pragma unroll
  for (int c = 0; c < 16; c += 8) {
    bf16x8 a, b;
    *reinterpret_cast<int4*>(&a) = *reinterpret_cast<const int4*>(
        dO + row * HS + c0 + c);
    *reinterpret_cast<int4*>(&b) = *reinterpret_cast<const int4*>(
        O + row * HS + c0 + c);
    #pragma unroll
    for (int i = 0; i < 8; ++i)
      s += (float)a[i] * (float)b[i];
  }
  #pragma unroll
  for (int w = 1; w <= 2; w <<= 1)
    s += __shfl_xor(s, w, 64);
  if ((threadIdx.x & 3) == 0) D[row] = s;
}

// dK/dV: one workgroup per KV tile; waves own 16 KV rows each.
__global__ __launch_bounds__(256) void flash_bwd_dkv_kernel(
    const __hip_bfloat16* __restrict__ Q,
    const __hip_bfloat16* __restrict__ K,
    const __hip_bfloat16* __restrict__ V,
    const __hip_bfloat16* __restrict__ dOg,
    const float* __restrict__ LSE, const float* __restrict__ Dg,
    __hip_bfloat16* __restrict__ dK, __hip_bfloat16* __restrict__ dV,
    int T, float scale) {
  const int bh = blockIdx.y;
  const int kn0 = blockIdx.x * BN;
  if (kn0 >= T) return;
  const long base = (long)bh * T * HS;
  const __hip_bfloat16* q = Q + base;
  const __hip_bfloat16* k = K + base;
  const __hip_bfloat16* v = V + base;
  const __hip_bfloat16* dO = dOg + base;
  const float* lse = LSE + (long)bh * T;
  const float* Drow = Dg + (long)bh * T;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wkv0 = kn0 + wave * 16;  // wave's first KV row

  __shared__ __bf16 lds_dot[HS * BN];   // dO^T image [hs][q]
  __shared__ __bf16 lds_qt[HS * BN];    // Q^T image  [hs][q]
  __shared__ __bf16 lds_pt[4][16 * BN]; // per-wave P^T [16][q]
  __shared__ __bf16 lds_dst[4][16 * BN];// per-wave dS^T
  __shared__ float lds_lse[BN];
  __shared__ float lds_d[BN];

  // K and V fragments for this wave's rows (2 k-chunks over hs)
  bf16x8 kf[2], vf[2];
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    kf[kk] = load_frag_rowmajor(k, wkv0, kk, lane, HS);
    vf[kk] = load_frag_rowmajor(v, wkv0, kk, lane, HS);
  }

  f32x4 dv_acc[4], dk_acc[4];
  #pragma unroll
  for (int n = 0; n < 4; ++n) {
    dv_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
    dk_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const float l2e = 1.4426950408889634f;
  const int col0 = lane & 15;

  for (int qm0 = kn0; qm0 < T; qm0 += BM) {
    // stage dO^T and Q^T images + lse/D rows (cooperative)
    {
      const int r = threadIdx.x >> 3;
      const int c0 = (threadIdx.x & 7) * 8;
      #pragma unroll
      for (int rep = 0; rep < 2; ++rep) {
        const int row = r + rep * 32;
        bf16x8 t1, t2;
        *reinterpret_cast<int4*>(&t1) = *reinterpret_cast<const int4*>(
            dO + (long)(qm0 + row) * HS + c0);
        *reinterpret_cast<int4*>(&t2) = *reinterpret_cast<const int4*>(
            q + (long)(qm0 + row) * HS + c0);
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
          lds_dot[(c0 + i) * BN + row] = t1[i];
          lds_qt[(c0 + i) * BN + row] = t2[i];
        }
      }
      if (threadIdx.x < BN) {
        lds_lse[threadIdx.x] = lse[qm0 + threadIdx.x];
        lds_d[threadIdx.x] = Drow[qm0 + threadIdx.x];
      }
    }
    __syncthreads();

    // S^T = K · Q^T  -> [16 kv, 64 q] per wave
    f32x4 st[4];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      st[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        // B = Q^T: col = q row (16n + col0), k = hs -> direct from Q
        bf16x8 bq = load_frag_rowmajor(q, qm0 + 16 * n, kk, lane, HS);
        st[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            kf[kk], bq, st[n], 0, 0, 0);
      }
    }
    // dP^T = V · dO^T
    f32x4 dpt[4];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      dpt[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 bd = load_frag_rowmajor(dO, qm0 + 16 * n, kk, lane, HS);
        dpt[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            vf[kk], bd, dpt[n], 0, 0, 0);
      }
    }

    // P^T = exp(scale*S^T - lse[q]); dS^T = scale * P^T*(dP^T - D[q])
    __bf16* pt = lds_pt[wave];
    __bf16* dst = lds_dst[wave];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int qcol = 16 * n + col0;
      const float lse_q = lds_lse[qcol];
      const float d_q = lds_d[qcol];
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvrow = wkv0 + (lane >> 4) * 4 + r;
        const bool valid = (qm0 + qcol) >= kvrow;  // causal
        const float p = valid
            ? exp2f((scale * st[n][r] - lse_q) * l2e) : 0.f;
        const float ds = scale * p * (dpt[n][r] - d_q);
        pt[((lane >> 4) * 4 + r) * BN + qcol] = (__bf16)p;
        dst[((lane >> 4) * 4 + r) * BN + qcol] =
            (__bf16)(valid ? ds : 0.f);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);

    // dV += P^T · dO  (B needs dO^T image);  dK += dS^T · Q (Q^T image)
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 pa, da;
      *reinterpret_cast<int4*>(&pa) = *reinterpret_cast<const int4*>(
          pt + (lane & 15) * BN + kk * 32 + (lane >> 4) * 8);
      *reinterpret_cast<int4*>(&da) = *reinterpret_cast<const int4*>(
          dst + (lane & 15) * BN + kk * 32 + (lane >> 4) * 8);
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 bdo, bq;
        // B[k=q][n=hs] from the transposed images: row = hs col
        *reinterpret_cast<int4*>(&bdo) = *reinterpret_cast<const int4*>(
            lds_dot + (16 * n + (lane & 15)) * BN + kk * 32 +
            (lane >> 4) * 8);
        *reinterpret_cast<int4*>(&bq) = *reinterpret_cast<const int4*>(
            lds_qt + (16 * n + (lane & 15)) * BN + kk * 32 +
            (lane >> 4) * 8);
        dv_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, bdo, dv_acc[n], 0, 0, 0);
        dk_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            da, bq, dk_acc[n], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: store dK, dV (C layout -> row-major)
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvrow = wkv0 + (lane >> 4) * 4 + r;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      dK[base + (long)kvrow * HS + 16 * n + col0] =
          __float2bfloat16(dk_acc[n][r]);
      dV[base + (long)kvrow * HS + 16 * n + col0] =
          __float2bfloat16(dv_acc[n][r]);
    }
  }
}

// dQ: one workgroup per Q tile; waves own 16 query rows.
__global__ __launch_bounds__(256) void flash_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ Q,
    const __hip_bfloat16* __restrict__ K,
    const __hip_bfloat16* __restrict__ V,
    const __hip_bfloat16* __restrict__ dOg,
    const float* __restrict__ LSE, const float* __restrict__ Dg,
    __hip_bfloat16* __restrict__ dQ, int T, float scale) {
  const int bh = blockIdx.y;
  const int qm0 = blockIdx.x * BM;
  if (qm0 >= T) return;
  const long base = (long)bh * T * HS;
  const __hip_bfloat16* q = Q + base;
  const __hip_bfloat16* k = K + base;
  const __hip_bfloat16* v = V + base;
  const __hip_bfloat16* dO = dOg + base;
  const float* lse = LSE + (long)bh * T;
  const float* Drow = Dg + (long)bh * T;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow0 = qm0 + wave * 16;
  const int col0 = lane & 15;

  __shared__ __bf16 lds_kt[HS * BN];    // K^T [hs][kv]
  __shared__ __bf16 lds_ds[4][16 * BN]; // per-wave dS

  bf16x8 qf[2], dof[2];
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    qf[kk] = load_frag_rowmajor(q, wrow0, kk, lane, HS);
    dof[kk] = load_frag_rowmajor(dO, wrow0, kk, lane, HS);
  }
  float lse_r[4], d_r[4];
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = wrow0 + (lane >> 4) * 4 + r;
    lse_r[r] = lse[qrow];
    d_r[r] = Drow[qrow];
  }

  f32x4 dq_acc[4];
  #pragma unroll
  for (int n = 0; n < 4; ++n) dq_acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const float l2e = 1.4426950408889634f;
  const int kv_end = qm0 + BM;
  for (int kn0 = 0; kn0 < kv_end && kn0 < T; kn0 += BN) {
    // stage the K^T image (dQ's B operand contracts over kv)
    {
      const int r = threadIdx.x >> 3;
      const int c0 = (threadIdx.x & 7) * 8;
      #pragma unroll
      for (int rep = 0; rep < 2; ++rep) {
        const int row = r + rep * 32;
        bf16x8 t2;
        *reinterpret_cast<int4*>(&t2) = *reinterpret_cast<const int4*>(
            k + (long)(kn0 + row) * HS + c0);
        #pragma unroll
        for (int i = 0; i < 8; ++i)
          lds_kt[(c0 + i) * BN + row] = t2[i];
      }
    }
    __syncthreads();

    // S = Q K^T (direct K frags), dP = dO V^T (V^T image)
    f32x4 s[4], dp[4];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      s[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      dp[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 bk = load_frag_rowmajor(k, kn0 + 16 * n, kk, lane, HS);
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qf[kk], bk, s[n], 0, 0, 0);
        // dP = dO · V^T contracts over hs -> row-major V fragments
        // (col = kv row, k = hs), like the S kernel's K fragments
        bf16x8 bv = load_frag_rowmajor(v, kn0 + 16 * n, kk, lane, HS);
        dp[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dof[kk], bv, dp[n], 0, 0, 0);
      }
    }

    // dS = scale * P * (dP - D)
    __bf16* dsw = lds_ds[wave];
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = wrow0 + (lane >> 4) * 4 + r;
        const int key = kn0 + 16 * n + col0;
        const bool valid = key <= qrow;
        const float p = valid
            ? exp2f((scale * s[n][r] - lse_r[r]) * l2e) : 0.f;
        const float ds = scale * p * (dp[n][r] - d_r[r]);
        dsw[((lane >> 4) * 4 + r) * BN + 16 * n + col0] =
            (__bf16)(valid ? ds : 0.f);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);

    // dQ += dS · K  (B = K^T image: B[k=kv][n=hs])
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 dsa;
      *reinterpret_cast<int4*>(&dsa) = *reinterpret_cast<const int4*>(
          dsw + (lane & 15) * BN + kk * 32 + (lane >> 4) * 8);
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 bk;
        *reinterpret_cast<int4*>(&bk) = *reinterpret_cast<const int4*>(
            lds_kt + (16 * n + col0) * BN + kk * 32 + (lane >> 4) * 8);
        dq_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, bk, dq_acc[n], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = wrow0 + (lane >> 4) * 4 + r;
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      dQ[base + (long)qrow * HS + 16 * n + col0] =
          __float2bfloat16(dq_acc[n][r]);
  }
}

}  // namespace

std::vector<torch::Tensor> flash_attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k,
    torch::Tensor v, torch::Tensor o, torch::Tensor lse, double scale) {
  const int B = (int)q.size(0), H = (int)q.size(1), T = (int)q.size(2);
  dout = dout.contiguous();
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto D = torch::empty({B, H, T}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  const long total_rows = (long)B * H * T;
  hipLaunchKernelGGL(flash_bwd_pre_kernel,
                     dim3((total_rows + 63) / 64), dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(o.data_ptr()),
                     D.data_ptr<float>(), total_rows);
  dim3 grid(T / BN, B * H);
  hipLaunchKernelGGL(flash_bwd_dkv_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr()),
                     lse.data_ptr<float>(), D.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()),
                     T, (float)scale);
  hipLaunchKernelGGL(flash_bwd_dq_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr()),
                     lse.data_ptr<float>(), D.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),
                     T, (float)scale);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "flash_bwd: ", hipGetErrorString(e));
  return {dq, dk, dv};
}
