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
