// MFMA fragment-layout probe for gfx950 (cdna4_isa.md is not shipped in
// this environment, so the lane->element maps used by flash_attn.hip
// are verified empirically: tests/test_flash_attn_gpu.py checks this
// kernel's 16x16 output against torch.matmul with ASYMMETRIC operands
// per the guide's A=I-check rule).
//
// Assumed maps (CDNA3 pattern extended to gfx950's 2xK):
//   mfma_f32_16x16x32_bf16:
//     A[16M x 32K]: lane l, elem i (of 8) -> row = l & 15,
//                                            k   = (l >> 4) * 8 + i
//     B[32K x 16N]: lane l, elem i        -> col = l & 15,
//                                            k   = (l >> 4) * 8 + i
//     C/D (4 f32): col = l & 15, row = (l >> 4) * 4 + reg   (guide §3)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void mfma_probe_16x16x32(const __hip_bfloat16* __restrict__ A,
                                    const __hip_bfloat16* __restrict__ B,
                                    float* __restrict__ C) {
  // one wave; A is [16, 32] row-major, B is [32, 16] row-major
  const int l = threadIdx.x;
  if (l >= 64) return;
  bf16x8 a, b;
  const int arow = l & 15, kgrp = (l >> 4) * 8;
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = *reinterpret_cast<const __bf16*>(&A[arow * 32 + kgrp + i]);
    b[i] = *reinterpret_cast<const __bf16*>(&B[(kgrp + i) * 16 + (l & 15)]);
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = (l >> 4) * 4 + r;
    const int col = l & 15;
    C[row * 16 + col] = c[r];
  }
}

}  // namespace

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) &&
              B.sizes() == torch::IntArrayRef({32, 16}) &&
              A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16);
  auto C = torch::zeros({16, 16},
                        A.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(mfma_probe_16x16x32, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                     C.data_ptr<float>());
  return C;
}

// ---------------------------------------------------------------------
// perm_probe: verifies the in-register C-layout -> A-fragment
// redistribution (ds_bpermute + v_perm) that the round-2 flash
// redesign needs to eliminate the P LDS round-trip
// (profiles/r01_flash_attn_notes.md).
//
// A wave holds M[16][64] f32 in MFMA C layout (lane l owns rows
// (l>>4)*4+r, col l&15 + 16n). The probe packs row-pairs with
// v_cvt_pk_bf16_f32 (compiler form), redistributes to the A-fragment
// layout (row l&15, k (l>>4)*8+i) with 2 bpermutes + 1 byte-perm per
// output dword, runs C2 = M x B through mfma_16x16x32, and the test
// compares against torch.matmul.
// ---------------------------------------------------------------------
namespace {

using bf16x8_p = __attribute__((ext_vector_type(8))) __bf16;
using f32x4_p = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ unsigned pack_bf16_pair(float lo, float hi) {
  __bf16 l = (__bf16)lo, h = (__bf16)hi;
  unsigned short ul = __builtin_bit_cast(unsigned short, l);
  unsigned short uh = __builtin_bit_cast(unsigned short, h);
  return (unsigned)ul | ((unsigned)uh << 16);
}

__global__ void perm_probe_kernel(const float* __restrict__ M,
                                  const __hip_bfloat16* __restrict__ B,
                                  float* __restrict__ C2, int variant) {
  const int l = threadIdx.x;
  if (l >= 64) return;
  const int g = l >> 4;        // lane group
  const int c15 = l & 15;

  // 1. load M into the C layout this lane would hold after an MFMA
  float s[4][4];  // [r][n]: element (row=(g*4+r), col=c15+16n)
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      s[r][n] = M[(g * 4 + r) * 64 + c15 + 16 * n];

  // 2. pack row-pairs: p[q][n] = bf16(row 4g+2q, col) | bf16(row
  //    4g+2q+1, col) << 16
  unsigned p[2][4];
  #pragma unroll
  for (int q = 0; q < 2; ++q)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      p[q][n] = pack_bf16_pair(s[2 * q][n], s[2 * q + 1][n]);

  // 3. redistribute to A-fragment layout: target lane l needs
  //    A[row=c15][k=kk*32+g*8+i]; source of element (row, col):
  //    lane (row>>2)*16 + (col&15), reg q=(row&3)>>1, n=col>>4,
  //    half = row&1.
  const int q_t = (c15 & 3) >> 1;          // fixed per target lane
  const int parity = c15 & 1;
  // perm byte pool: {second operand bytes 0-3, first operand bytes
  // 4-7} per llvm.amdgcn.perm; variant 1 assumes the opposite order —
  // the host test reports which matches (both computed in one pass,
  // selected by the `variant` flag).
  const unsigned sel0 = parity ? 0x07060302u : 0x05040100u;
  const unsigned sel1 = parity ? 0x03020706u : 0x01000504u;
  bf16x8_p afrag[2];
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int n_t = (kk * 32 + g * 8) >> 4;  // fixed per (lane, kk)
    unsigned out[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col_e = kk * 32 + g * 8 + 2 * j;
      const int lane_e = (c15 >> 2) * 16 + (col_e & 15);
      // ds_bpermute wants byte addresses (lane*4)
      unsigned A = __builtin_amdgcn_ds_bpermute(lane_e * 4,
                                                p[q_t][n_t]);
      unsigned Bv = __builtin_amdgcn_ds_bpermute((lane_e + 1) * 4,
                                                 p[q_t][n_t]);
      out[j] = variant
          ? __builtin_amdgcn_perm(Bv, A, sel1)
          : __builtin_amdgcn_perm(Bv, A, sel0);
    }
    afrag[kk] = __builtin_bit_cast(bf16x8_p,
        (__attribute__((ext_vector_type(4))) unsigned){
            out[0], out[1], out[2], out[3]});
  }

  // 4. MFMA against B [64, 16] and write C2 = M x B
  f32x4_p acc = {0.f, 0.f, 0.f, 0.f};
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    bf16x8_p bf;
    #pragma unroll
    for (int i = 0; i < 8; ++i)
      bf[i] = *reinterpret_cast<const __bf16*>(
          &B[(kk * 32 + g * 8 + i) * 16 + c15]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kk], bf, acc,
                                                  0, 0, 0);
  }
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    C2[(g * 4 + r) * 16 + c15] = acc[r];
}

}  // namespace

torch::Tensor perm_probe(torch::Tensor M, torch::Tensor B,
                         long variant) {
  TORCH_CHECK(M.sizes() == torch::IntArrayRef({16, 64}) &&
              M.scalar_type() == at::kFloat &&
              B.sizes() == torch::IntArrayRef({64, 16}) &&
              B.scalar_type() == at::kBFloat16);
  auto C2 = torch::zeros({16, 16}, M.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(perm_probe_kernel, dim3(1), dim3(64), 0, stream,
                     M.data_ptr<float>(),
                     reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                     C2.data_ptr<float>(), (int)variant);
  return C2;
}

// Diagnostic: dump the redistributed A-fragment (and a raw bpermute
// identity check) so a failing perm_probe pinpoints the wrong stage.
namespace {
__global__ void perm_dump_kernel(const float* __restrict__ M,
                                 float* __restrict__ out,
                                 unsigned* __restrict__ ident,
                                 int variant) {
  const int l = threadIdx.x;
  if (l >= 64) return;
  const int g = l >> 4;
  const int c15 = l & 15;
  // bpermute identity: lane l fetches lane (l+1)%64's value l+100
  ident[l] = __builtin_amdgcn_ds_bpermute(((l + 1) % 64) * 4,
                                          (unsigned)(l + 100));
  float s[4][4];
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      s[r][n] = M[(g * 4 + r) * 64 + c15 + 16 * n];
  unsigned p[2][4];
  #pragma unroll
  for (int q = 0; q < 2; ++q)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      p[q][n] = pack_bf16_pair(s[2 * q][n], s[2 * q + 1][n]);
  const int q_t = (c15 & 3) >> 1;
  const int parity = c15 & 1;
  const unsigned sel0 = parity ? 0x07060302u : 0x05040100u;
  const unsigned sel1 = parity ? 0x03020706u : 0x01000504u;
  #pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int n_t = (kk * 32 + g * 8) >> 4;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col_e = kk * 32 + g * 8 + 2 * j;
      const int lane_e = (c15 >> 2) * 16 + (col_e & 15);
      unsigned A = __builtin_amdgcn_ds_bpermute(lane_e * 4,
                                                p[q_t][n_t]);
      unsigned Bv = __builtin_amdgcn_ds_bpermute((lane_e + 1) * 4,
                                                 p[q_t][n_t]);
      unsigned o = variant ? __builtin_amdgcn_perm(Bv, A, sel1)
                           : __builtin_amdgcn_perm(Bv, A, sel0);
      __bf16 lo = __builtin_bit_cast(__bf16,
          (unsigned short)(o & 0xffff));
      __bf16 hi = __builtin_bit_cast(__bf16,
          (unsigned short)(o >> 16));
      // element (row=c15, k=col_e) and (row=c15, k=col_e+1)
      out[(long)c15 * 64 + col_e] = (float)lo;
      out[(long)c15 * 64 + col_e + 1] = (float)hi;
    }
  }
}
}  // namespace

std::vector<torch::Tensor> perm_dump(torch::Tensor M, long variant) {
  auto out = torch::zeros({16, 64}, M.options());
  auto ident = torch::zeros({64},
                            M.options().dtype(at::kInt));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(perm_dump_kernel, dim3(1), dim3(64), 0, stream,
                     M.data_ptr<float>(), out.data_ptr<float>(),
                     reinterpret_cast<unsigned*>(ident.data_ptr()),
                     (int)variant);
  return {out, ident};
}

namespace {
// permlane{16,32}_swap semantics probe: a = 0x1000+lane, b = 0x2000+lane.
// Row 0/1: permlane32_swap results (r0, r1); row 2/3: permlane16_swap.
__global__ void permlane_swap_probe_kernel(unsigned* __restrict__ out) {
  const unsigned lane = threadIdx.x;
  if (lane >= 64) return;
  const unsigned a = 0x1000u + lane;
  const unsigned b = 0x2000u + lane;
  {
    auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
    out[lane] = r[0];
    out[64 + lane] = r[1];
  }
  {
    auto r = __builtin_amdgcn_permlane16_swap(a, b, false, false);
    out[128 + lane] = r[0];
    out[192 + lane] = r[1];
  }
}
}  // namespace

torch::Tensor permlane_swap_probe() {
  auto out = torch::zeros({4, 64}, torch::dtype(at::kInt)
                                       .device(at::kCUDA));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(permlane_swap_probe_kernel, dim3(1), dim3(64), 0,
                     stream,
                     reinterpret_cast<unsigned*>(out.data_ptr()));
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "permlane_swap_probe: ",
              hipGetErrorString(e));
  return out;
}

namespace {
// ds_read_b64_tr_b16 semantics probe: LDS u16[4096] filled with its own
// index; each lane reads 8 B at addr = a*lane + b*(lane&15) + c*(lane>>4)
// and we dump the 4 u16 element-indices each lane received.
__global__ void tr16_probe_kernel(unsigned short* __restrict__ out,
                                  int a, int b, int c) {
  __shared__ unsigned short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x)
    lds[i] = (unsigned short)i;
  __syncthreads();
  if (threadIdx.x >= 64) return;
  const int lane = threadIdx.x;
  const int addr = a * lane + b * (lane & 15) + c * (lane >> 4);
  // AS(3) pointer -> 32-bit LDS byte offset for the ds instruction
  typedef __attribute__((ext_vector_type(2))) unsigned uint2v;
  const unsigned base =
      (unsigned)(size_t)(__attribute__((address_space(3))) char*)
          (void*)lds;
  uint2v r;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %1\n"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r)
      : "v"(base + (unsigned)addr)
      : "memory");
  out[lane * 4 + 0] = (unsigned short)(r[0] & 0xffff);
  out[lane * 4 + 1] = (unsigned short)(r[0] >> 16);
  out[lane * 4 + 2] = (unsigned short)(r[1] & 0xffff);
  out[lane * 4 + 3] = (unsigned short)(r[1] >> 16);
}
}  // namespace

torch::Tensor tr16_probe(long a, long b, long c) {
  auto out = torch::zeros({64, 4}, torch::dtype(at::kShort)
                                       .device(at::kCUDA));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(256), 0, stream,
                     reinterpret_cast<unsigned short*>(out.data_ptr()),
                     (int)a, (int)b, (int)c);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "tr16_probe: ", hipGetErrorString(e));
  return out;
}

namespace {
using f32x16 = __attribute__((ext_vector_type(16))) float;

// 32x32x16 bf16 layout probe. Assumed maps (guide §3 + CDNA pattern):
//   A[32M x 16K]: lane l elem i -> row = l & 31, k = (l >> 5) * 8 + i
//   B[16K x 32N]: lane l elem i -> col = l & 31, k = (l >> 5) * 8 + i
//   C (16 f32):   col = l & 31, row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5)
__global__ void mfma_probe_32x32x16(const __hip_bfloat16* __restrict__ A,
                                    const __hip_bfloat16* __restrict__ B,
                                    float* __restrict__ C) {
  const int l = threadIdx.x;
  if (l >= 64) return;
  bf16x8 a, b;
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)(float)A[(l & 31) * 16 + (l >> 5) * 8 + i];
    b[i] = (__bf16)(float)B[((l >> 5) * 8 + i) * 32 + (l & 31)];
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    C[row * 32 + (l & 31)] = acc[r];
  }
}
}  // namespace

torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) &&
              B.sizes() == at::IntArrayRef({16, 32}));
  auto C = torch::zeros({32, 32},
                        A.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(mfma_probe_32x32x16, dim3(1), dim3(64), 0, stream,
      reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
      C.data_ptr<float>());
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "mfma_probe32: ", hipGetErrorString(e));
  return C;
}
