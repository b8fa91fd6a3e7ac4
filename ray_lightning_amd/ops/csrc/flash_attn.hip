// Flash attention (causal, hs=64, bf16) — hand-written CDNA4 MFMA
// kernels. AOTriton's SDPA kernels on gfx950 measure ~186 TF/s
// causal-effective forward and ~115 TF/s backward on the GPT-2 shapes
// (profiles/r01_gpt2xl_1gpu_baseline.md); these kernels use the
// guide's plain-HIP GEMM idioms (LDS-staged tiles, 16x16x32 MFMA,
// flash online softmax), 32 rows per wave (2 M-halves), per-wave
// skipping of fully-masked diagonal tiles.
//
// Layout: q, k, v are [B, H, T, 64] bf16 contiguous. Causal only.
// Fragment maps verified by mfma_probe (tests/test_flash_attn_gpu.py):
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
constexpr int WM = 32;      // query rows per wave (2 MFMA M-halves)
constexpr int BM = 128;     // query rows per workgroup (4 waves)
constexpr int BN = 64;      // key/value rows per tile
constexpr float kNegInf = -1e30f;

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
__device__ __forceinline__ void wait_lds() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

// 16 bytes per lane from a [rows, HS] row-major bf16 tile:
// row = base_row + (l&15), k = kk*32 + (l>>4)*8 .. +8
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

// cooperative transpose-stage of a [BN, HS] tile into dst[HS][BN]
__device__ __forceinline__ void stage_transposed(
    __bf16* dst, const __hip_bfloat16* src, int row0) {
  const int r = threadIdx.x >> 3;
  const int c0 = (threadIdx.x & 7) * 8;
  #pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    const int row = r + rep * 32;
    bf16x8 t;
    *reinterpret_cast<int4*>(&t) = *reinterpret_cast<const int4*>(
        src + (long)(row0 + row) * HS + c0);
    #pragma unroll
    for (int i = 0; i < 8; ++i)
      dst[(c0 + i) * BN + row] = t[i];
  }
}

// ---------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void flash_fwd_kernel(
    const __hip_bfloat16* __restrict__ Q,
    const __hip_bfloat16* __restrict__ K,
    const __hip_bfloat16* __restrict__ V,
    __hip_bfloat16* __restrict__ O, float* __restrict__ LSE,
    int T, float scale) {
  const int bh = blockIdx.y;
  const int qm0 = blockIdx.x * BM;
  if (qm0 >= T) return;
  const long base = (long)bh * T * HS;
  const __hip_bfloat16* q = Q + base;
  const __hip_bfloat16* k = K + base;
  const __hip_bfloat16* v = V + base;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow0 = qm0 + wave * WM;   // wave's first query row
  const int col0 = lane & 15;

  // V^T double-buffered: tile i writes buf[i&1] while others still
  // read buf[(i-1)&1] -> ONE barrier per tile instead of two
  __shared__ __bf16 lds_vt[2][HS * BN];       // V^T [hs][kv] 16 KB
  __shared__ __bf16 lds_p[4][WM * BN];        // per-wave P   16 KB

  bf16x8 qf[2][2];  // [m-half][k-chunk]
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      qf[mh][kk] = load_frag_rowmajor(q, wrow0 + mh * 16, kk, lane, HS);

  float m_i[2][4], l_i[2][4];
  f32x4 o_acc[2][4];
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_i[mh][r] = kNegInf;
      l_i[mh][r] = 0.f;
    }
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      o_acc[mh][n] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const float l2e = 1.4426950408889634f * scale;
  const int kv_end = qm0 + BM;
  int buf = 0;
  for (int kn0 = 0; kn0 < kv_end && kn0 < T; kn0 += BN, buf ^= 1) {
    stage_transposed(lds_vt[buf], v, kn0);
    __syncthreads();
    // waves whose every query row is below this KV tile skip compute
    const bool active = kn0 <= wrow0 + WM - 1;
    if (active) {
      // --- S = Q K^T ------------------------------------------------
      f32x4 s_acc[2][4];
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 bf0 = load_frag_rowmajor(k, kn0 + 16 * n, 0, lane, HS);
        bf16x8 bf1 = load_frag_rowmajor(k, kn0 + 16 * n, 1, lane, HS);
        #pragma unroll
        for (int mh = 0; mh < 2; ++mh) {
          s_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mh][0], bf0, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
          s_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mh][1], bf1, s_acc[mh][n], 0, 0, 0);
        }
      }
      // --- causal mask (diagonal tiles only) ------------------------
      if (kn0 + BN > wrow0) {
        #pragma unroll
        for (int mh = 0; mh < 2; ++mh)
          #pragma unroll
          for (int n = 0; n < 4; ++n)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int qrow = wrow0 + mh * 16 + (lane >> 4) * 4 + r;
              if (kn0 + 16 * n + col0 > qrow)
                s_acc[mh][n][r] = kNegInf;
            }
      }
      // --- online softmax -------------------------------------------
      #pragma unroll
      for (int mh = 0; mh < 2; ++mh) {
        float p_scale[4];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          float mx = fmaxf(fmaxf(s_acc[mh][0][r], s_acc[mh][1][r]),
                           fmaxf(s_acc[mh][2][r], s_acc[mh][3][r]));
          mx = row_reduce_max(mx);
          const float m_new = fmaxf(m_i[mh][r], mx);
          const float corr = exp2f((m_i[mh][r] - m_new) * l2e);
          float rowsum = 0.f;
          #pragma unroll
          for (int n = 0; n < 4; ++n) {
            const float p = (s_acc[mh][n][r] <= kNegInf * 0.5f)
                ? 0.f : exp2f((s_acc[mh][n][r] - m_new) * l2e);
            s_acc[mh][n][r] = p;
            rowsum += p;
          }
          rowsum = row_reduce_sum(rowsum);
          l_i[mh][r] = l_i[mh][r] * corr + rowsum;
          m_i[mh][r] = m_new;
          p_scale[r] = corr;
        }
        #pragma unroll
        for (int n = 0; n < 4; ++n)
          #pragma unroll
          for (int r = 0; r < 4; ++r)
            o_acc[mh][n][r] *= p_scale[r];
      }
      // --- stage P --------------------------------------------------
      __bf16* pw = lds_p[wave];
      #pragma unroll
      for (int mh = 0; mh < 2; ++mh)
        #pragma unroll
        for (int n = 0; n < 4; ++n)
          #pragma unroll
          for (int r = 0; r < 4; ++r)
            pw[(mh * 16 + (lane >> 4) * 4 + r) * BN + 16 * n + col0] =
                (__bf16)s_acc[mh][n][r];
      wait_lds();
      // --- O += P V -------------------------------------------------
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 pa[2];
        #pragma unroll
        for (int mh = 0; mh < 2; ++mh)
          *reinterpret_cast<int4*>(&pa[mh]) =
              *reinterpret_cast<const int4*>(
                  pw + (mh * 16 + (lane & 15)) * BN + kk * 32 +
                  (lane >> 4) * 8);
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          bf16x8 vb;
          *reinterpret_cast<int4*>(&vb) =
              *reinterpret_cast<const int4*>(
                  lds_vt[buf] + (16 * n + col0) * BN + kk * 32 +
                  (lane >> 4) * 8);
          #pragma unroll
          for (int mh = 0; mh < 2; ++mh)
            o_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pa[mh], vb, o_acc[mh][n], 0, 0, 0);
        }
      }
    }
    // no trailing barrier: next tile writes the OTHER V^T buffer
  }

  // --- epilogue -----------------------------------------------------
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = wrow0 + mh * 16 + (lane >> 4) * 4 + r;
      const float inv_l = (l_i[mh][r] > 0.f) ? 1.f / l_i[mh][r] : 0.f;
      #pragma unroll
      for (int n = 0; n < 4; ++n)
        O[base + (long)qrow * HS + 16 * n + col0] =
            __float2bfloat16(o_acc[mh][n][r] * inv_l);
      if (col0 == 0)
        LSE[(long)bh * T + qrow] =
            m_i[mh][r] * scale + logf(fmaxf(l_i[mh][r], 1e-30f));
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
  TORCH_CHECK(T % BM == 0, "flash_attn_fwd: T % 128 == 0");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, T}, q.options().dtype(at::kFloat));
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

// D = rowsum(dO * O), fp32
__global__ __launch_bounds__(256) void flash_bwd_pre_kernel(
    const __hip_bfloat16* __restrict__ dO,
    const __hip_bfloat16* __restrict__ O, float* __restrict__ D,
    long total_rows) {
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

// dK/dV: one workgroup per 128 KV rows; waves own 32 KV rows
__global__ __launch_bounds__(256) void flash_bwd_dkv_kernel(
    const __hip_bfloat16* __restrict__ Q,
    const __hip_bfloat16* __restrict__ K,
    const __hip_bfloat16* __restrict__ V,
    const __hip_bfloat16* __restrict__ dOg,
    const float* __restrict__ LSE, const float* __restrict__ Dg,
    __hip_bfloat16* __restrict__ dK, __hip_bfloat16* __restrict__ dV,
    int T, float scale) {
  const int bh = blockIdx.y;
  const int kb0 = blockIdx.x * BM;   // 128 KV rows per block
  if (kb0 >= T) return;
  const long base = (long)bh * T * HS;
  const __hip_bfloat16* q = Q + base;
  const __hip_bfloat16* k = K + base;
  const __hip_bfloat16* v = V + base;
  const __hip_bfloat16* dO = dOg + base;
  const float* lse = LSE + (long)bh * T;
  const float* Drow = Dg + (long)bh * T;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wkv0 = kb0 + wave * WM;   // wave's first KV row
  const int col0 = lane & 15;

  __shared__ __bf16 lds_dot[HS * BN];     // dO^T [hs][q]   8 KB
  __shared__ __bf16 lds_qt[HS * BN];      // Q^T  [hs][q]   8 KB
  __shared__ __bf16 lds_pt[4][WM * BN];   // P^T  per wave 16 KB
  __shared__ __bf16 lds_dst[4][WM * BN];  // dS^T per wave 16 KB
  __shared__ float lds_lse[BN];
  __shared__ float lds_d[BN];

  bf16x8 kf[2][2], vf[2][2];  // [kv-half][k-chunk]
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      kf[mh][kk] = load_frag_rowmajor(k, wkv0 + mh * 16, kk, lane, HS);
      vf[mh][kk] = load_frag_rowmajor(v, wkv0 + mh * 16, kk, lane, HS);
    }

  f32x4 dv_acc[2][4], dk_acc[2][4];
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      dv_acc[mh][n] = f32x4{0.f, 0.f, 0.f, 0.f};
      dk_acc[mh][n] = f32x4{0.f, 0.f, 0.f, 0.f};
    }

  const float l2e = 1.4426950408889634f;

  for (int qm0 = kb0; qm0 < T; qm0 += BN) {  // q tiles of 64 rows
    stage_transposed(lds_dot, dO, qm0);
    stage_transposed(lds_qt, q, qm0);
    if (threadIdx.x < BN) {
      lds_lse[threadIdx.x] = lse[qm0 + threadIdx.x];
      lds_d[threadIdx.x] = Drow[qm0 + threadIdx.x];
    }
    __syncthreads();
    // skip when every q row here is below the wave's first KV row
    const bool active = wkv0 <= qm0 + BN - 1;
    if (active) {
      __bf16* pt = lds_pt[wave];
      __bf16* dst = lds_dst[wave];
      #pragma unroll
      for (int mh = 0; mh < 2; ++mh) {
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          // S^T = K Q^T ; dP^T = V dO^T   [16 kv, 16 q] tiles
          bf16x8 bq0 = load_frag_rowmajor(q, qm0 + 16 * n, 0, lane, HS);
          bf16x8 bq1 = load_frag_rowmajor(q, qm0 + 16 * n, 1, lane, HS);
          bf16x8 bd0 = load_frag_rowmajor(dO, qm0 + 16 * n, 0, lane,
                                          HS);
          bf16x8 bd1 = load_frag_rowmajor(dO, qm0 + 16 * n, 1, lane,
                                          HS);
          f32x4 st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kf[mh][0], bq0, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
          st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kf[mh][1], bq1, st, 0, 0, 0);
          f32x4 dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              vf[mh][0], bd0, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
          dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              vf[mh][1], bd1, dpt, 0, 0, 0);
          const int qcol = 16 * n + col0;
          const float lse_q = lds_lse[qcol];
          const float d_q = lds_d[qcol];
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int kvrow = wkv0 + mh * 16 + (lane >> 4) * 4 + r;
            const bool valid = (qm0 + qcol) >= kvrow;
            const float p = valid
                ? exp2f((scale * st[r] - lse_q) * l2e) : 0.f;
            const float ds = valid
                ? scale * p * (dpt[r] - d_q) : 0.f;
            pt[(mh * 16 + (lane >> 4) * 4 + r) * BN + qcol] =
                (__bf16)p;
            dst[(mh * 16 + (lane >> 4) * 4 + r) * BN + qcol] =
                (__bf16)ds;
          }
        }
      }
      wait_lds();
      // dV += P^T dO (dO^T image); dK += dS^T Q (Q^T image)
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 pa[2], da[2];
        #pragma unroll
        for (int mh = 0; mh < 2; ++mh) {
          *reinterpret_cast<int4*>(&pa[mh]) =
              *reinterpret_cast<const int4*>(
                  pt + (mh * 16 + (lane & 15)) * BN + kk * 32 +
                  (lane >> 4) * 8);
          *reinterpret_cast<int4*>(&da[mh]) =
              *reinterpret_cast<const int4*>(
                  dst + (mh * 16 + (lane & 15)) * BN + kk * 32 +
                  (lane >> 4) * 8);
        }
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          bf16x8 bdo, bq;
          *reinterpret_cast<int4*>(&bdo) =
              *reinterpret_cast<const int4*>(
                  lds_dot + (16 * n + col0) * BN + kk * 32 +
                  (lane >> 4) * 8);
          *reinterpret_cast<int4*>(&bq) =
              *reinterpret_cast<const int4*>(
                  lds_qt + (16 * n + col0) * BN + kk * 32 +
                  (lane >> 4) * 8);
          #pragma unroll
          for (int mh = 0; mh < 2; ++mh) {
            dv_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pa[mh], bdo, dv_acc[mh][n], 0, 0, 0);
            dk_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                da[mh], bq, dk_acc[mh][n], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int kvrow = wkv0 + mh * 16 + (lane >> 4) * 4 + r;
      #pragma unroll
      for (int n = 0; n < 4; ++n) {
        dK[base + (long)kvrow * HS + 16 * n + col0] =
            __float2bfloat16(dk_acc[mh][n][r]);
        dV[base + (long)kvrow * HS + 16 * n + col0] =
            __float2bfloat16(dv_acc[mh][n][r]);
      }
    }
}

// dQ: one workgroup per 128 query rows; waves own 32 rows
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
  const int wrow0 = qm0 + wave * WM;
  const int col0 = lane & 15;

  __shared__ __bf16 lds_kt[HS * BN];      // K^T [hs][kv]
  __shared__ __bf16 lds_ds[4][WM * BN];   // per-wave dS

  bf16x8 qf[2][2], dof[2][2];
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      qf[mh][kk] = load_frag_rowmajor(q, wrow0 + mh * 16, kk, lane, HS);
      dof[mh][kk] = load_frag_rowmajor(dO, wrow0 + mh * 16, kk, lane,
                                       HS);
    }
  float lse_r[2][4], d_r[2][4];
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = wrow0 + mh * 16 + (lane >> 4) * 4 + r;
      lse_r[mh][r] = lse[qrow];
      d_r[mh][r] = Drow[qrow];
    }

  f32x4 dq_acc[2][4];
  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      dq_acc[mh][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const float l2e = 1.4426950408889634f;
  const int kv_end = qm0 + BM;
  for (int kn0 = 0; kn0 < kv_end && kn0 < T; kn0 += BN) {
    {  // stage K^T image (dQ's B operand contracts over kv)
      const int r = threadIdx.x >> 3;
      const int c0 = (threadIdx.x & 7) * 8;
      #pragma unroll
      for (int rep = 0; rep < 2; ++rep) {
        const int row = r + rep * 32;
        bf16x8 t;
        *reinterpret_cast<int4*>(&t) = *reinterpret_cast<const int4*>(
            k + (long)(kn0 + row) * HS + c0);
        #pragma unroll
        for (int i = 0; i < 8; ++i)
          lds_kt[(c0 + i) * BN + row] = t[i];
      }
    }
    __syncthreads();
    const bool active = kn0 <= wrow0 + WM - 1;
    if (active) {
      __bf16* dsw = lds_ds[wave];
      #pragma unroll
      for (int mh = 0; mh < 2; ++mh) {
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          bf16x8 bk0 = load_frag_rowmajor(k, kn0 + 16 * n, 0, lane, HS);
          bf16x8 bk1 = load_frag_rowmajor(k, kn0 + 16 * n, 1, lane, HS);
          // dP = dO V^T contracts over hs -> row-major V fragments
          bf16x8 bv0 = load_frag_rowmajor(v, kn0 + 16 * n, 0, lane, HS);
          bf16x8 bv1 = load_frag_rowmajor(v, kn0 + 16 * n, 1, lane, HS);
          f32x4 s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mh][0], bk0, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
          s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mh][1], bk1, s, 0, 0, 0);
          f32x4 dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              dof[mh][0], bv0, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
          dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              dof[mh][1], bv1, dp, 0, 0, 0);
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int qrow = wrow0 + mh * 16 + (lane >> 4) * 4 + r;
            const int key = kn0 + 16 * n + col0;
            const bool valid = key <= qrow;
            const float p = valid
                ? exp2f((scale * s[r] - lse_r[mh][r]) * l2e) : 0.f;
            const float ds = valid
                ? scale * p * (dp[r] - d_r[mh][r]) : 0.f;
            dsw[(mh * 16 + (lane >> 4) * 4 + r) * BN + 16 * n + col0] =
                (__bf16)ds;
          }
        }
      }
      wait_lds();
      // dQ += dS K (K^T image)
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 dsa[2];
        #pragma unroll
        for (int mh = 0; mh < 2; ++mh)
          *reinterpret_cast<int4*>(&dsa[mh]) =
              *reinterpret_cast<const int4*>(
                  dsw + (mh * 16 + (lane & 15)) * BN + kk * 32 +
                  (lane >> 4) * 8);
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          bf16x8 bk;
          *reinterpret_cast<int4*>(&bk) =
              *reinterpret_cast<const int4*>(
                  lds_kt + (16 * n + col0) * BN + kk * 32 +
                  (lane >> 4) * 8);
          #pragma unroll
          for (int mh = 0; mh < 2; ++mh)
            dq_acc[mh][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                dsa[mh], bk, dq_acc[mh][n], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  #pragma unroll
  for (int mh = 0; mh < 2; ++mh)
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = wrow0 + mh * 16 + (lane >> 4) * 4 + r;
      #pragma unroll
      for (int n = 0; n < 4; ++n)
        dQ[base + (long)qrow * HS + 16 * n + col0] =
            __float2bfloat16(dq_acc[mh][n][r]);
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
  dim3 grid(T / BM, B * H);
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
