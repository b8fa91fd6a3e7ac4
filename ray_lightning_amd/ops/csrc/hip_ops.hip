// CDNA4 (gfx950) kernels for the framework-owned gradient/optimizer path.
//
// All kernels here are memory-bound: the design targets the HBM3E
// bandwidth ceiling (~6.3 TB/s achievable) with 16-byte-per-lane
// vectorized access (float4 / ushort8), 256-thread blocks (4 wave64),
// grids sized >> 256 workgroups to fill all 8 XCDs.
//
// Components (SURVEY.md N3/N6 hand-tuned halves):
//  - multi-tensor gradient pack (flatten + optional fp32<->bf16 cast)
//  - bucket scale / scale+cast (the 1/world_size unflatten step, fused)
//  - fused SGD(momentum) and Adam/AdamW multi-tensor optimizer steps
//  - sharded Adam step with bf16 params + fp32 master state
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <vector>

#define CHECK_HIP(cmd)                                                     \
  do {                                                                     \
    hipError_t e = (cmd);                                                  \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));     \
  } while (0)

static constexpr int kThreads = 256;   // 4 wave64
static constexpr int kMaxSeg = 48;     // segments per launch (kernarg cap)

// ---------------------------------------------------------------------
// dtype conversion helpers
// ---------------------------------------------------------------------
template <typename T> struct ToFloat;
template <> struct ToFloat<float> {
  static __device__ __forceinline__ float conv(float v) { return v; }
};
template <> struct ToFloat<__hip_bfloat16> {
  static __device__ __forceinline__ float conv(__hip_bfloat16 v) {
    return __bfloat162float(v);
  }
};
template <typename T> struct FromFloat;
template <> struct FromFloat<float> {
  static __device__ __forceinline__ float conv(float v) { return v; }
};
template <> struct FromFloat<__hip_bfloat16> {
  static __device__ __forceinline__ __hip_bfloat16 conv(float v) {
    return __float2bfloat16(v);
  }
};

// ---------------------------------------------------------------------
// multi-tensor pack: dst[i] = (DstT)src[i], segments are (src, dst,
// numel) with 16B-aligned dst (bucket offsets are 8-element aligned).
// ---------------------------------------------------------------------
template <typename SrcT, typename DstT>
struct PackMeta {
  const SrcT* src[kMaxSeg];
  DstT* dst[kMaxSeg];
  long numel[kMaxSeg];
  long start_block[kMaxSeg + 1];
  int n;
};

// elements each thread moves per iteration: 16 bytes of the *narrower*
// type so both sides use wide loads/stores
template <typename SrcT, typename DstT>
struct VecLen {
  static constexpr int value =
      16 / (sizeof(SrcT) > sizeof(DstT) ? sizeof(SrcT) : sizeof(DstT));
};

template <typename SrcT, typename DstT, int UNROLL = 4>
__global__ __launch_bounds__(kThreads) void multi_tensor_pack_kernel(
    PackMeta<SrcT, DstT> meta) {
  constexpr int V = VecLen<SrcT, DstT>::value;
  // find segment: linear scan is fine for <=48 entries (uniform branch)
  int seg = 0;
  long b = blockIdx.x;
  while (seg + 1 < meta.n && b >= meta.start_block[seg + 1]) seg++;
  const long local_block = b - meta.start_block[seg];
  const long numel = meta.numel[seg];
  const SrcT* __restrict__ src = meta.src[seg];
  DstT* __restrict__ dst = meta.dst[seg];

  const long block_elems = (long)kThreads * V * UNROLL;
  const long base = local_block * block_elems;
  #pragma unroll
  for (int u = 0; u < UNROLL; ++u) {
    const long idx = base + ((long)u * kThreads + threadIdx.x) * V;
    if (idx + V <= numel) {
      SrcT sv[V];
      DstT dv[V];
      // the wider side is 16B (int4), the narrower side 8B (int2)
      if constexpr (sizeof(SrcT) * V == 16)
        *reinterpret_cast<int4*>(sv) =
            *reinterpret_cast<const int4*>(src + idx);
      else
        *reinterpret_cast<int2*>(sv) =
            *reinterpret_cast<const int2*>(src + idx);
      #pragma unroll
      for (int i = 0; i < V; ++i)
        dv[i] = FromFloat<DstT>::conv(ToFloat<SrcT>::conv(sv[i]));
      if constexpr (sizeof(DstT) * V == 16)
        *reinterpret_cast<int4*>(dst + idx) =
            *reinterpret_cast<const int4*>(dv);
      else
        *reinterpret_cast<int2*>(dst + idx) =
            *reinterpret_cast<const int2*>(dv);
    } else {
      for (long i = idx; i < numel && i < idx + V; ++i)
        dst[i] = FromFloat<DstT>::conv(ToFloat<SrcT>::conv(src[i]));
    }
  }
}

template <typename SrcT, typename DstT>
static void launch_pack(const std::vector<torch::Tensor>& dsts,
                        const std::vector<torch::Tensor>& srcs,
                        hipStream_t stream) {
  constexpr int V = VecLen<SrcT, DstT>::value;
  constexpr int UNROLL = 4;
  const long block_elems = (long)kThreads * V * UNROLL;
  size_t i = 0;
  while (i < srcs.size()) {
    PackMeta<SrcT, DstT> meta;
    int n = 0;
    long blocks = 0;
    while (i < srcs.size() && n < kMaxSeg) {
      meta.src[n] = reinterpret_cast<const SrcT*>(srcs[i].data_ptr());
      meta.dst[n] = reinterpret_cast<DstT*>(dsts[i].data_ptr());
      meta.numel[n] = srcs[i].numel();
      meta.start_block[n] = blocks;
      blocks += (meta.numel[n] + block_elems - 1) / block_elems;
      ++n;
      ++i;
    }
    meta.start_block[n] = blocks;
    meta.n = n;
    if (blocks == 0) continue;
    hipLaunchKernelGGL((multi_tensor_pack_kernel<SrcT, DstT, UNROLL>),
                       dim3(blocks), dim3(kThreads), 0, stream, meta);
    CHECK_HIP(hipGetLastError());
  }
}

void multi_tensor_pack(std::vector<torch::Tensor> dsts,
                       std::vector<torch::Tensor> srcs) {
  TORCH_CHECK(dsts.size() == srcs.size(), "dst/src count mismatch");
  if (srcs.empty()) return;
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto sd = srcs[0].scalar_type();
  auto dd = dsts[0].scalar_type();
  for (size_t i = 0; i < srcs.size(); ++i) {
    TORCH_CHECK(srcs[i].is_contiguous() && dsts[i].is_contiguous(),
                "pack tensors must be contiguous");
    TORCH_CHECK(srcs[i].scalar_type() == sd &&
                dsts[i].scalar_type() == dd,
                "mixed dtypes in one pack call");
    TORCH_CHECK(srcs[i].numel() == dsts[i].numel(), "numel mismatch");
  }
  if (sd == at::kFloat && dd == at::kFloat)
    launch_pack<float, float>(dsts, srcs, stream);
  else if (sd == at::kFloat && dd == at::kBFloat16)
    launch_pack<float, __hip_bfloat16>(dsts, srcs, stream);
  else if (sd == at::kBFloat16 && dd == at::kBFloat16)
    launch_pack<__hip_bfloat16, __hip_bfloat16>(dsts, srcs, stream);
  else if (sd == at::kBFloat16 && dd == at::kFloat)
    launch_pack<__hip_bfloat16, float>(dsts, srcs, stream);
  else
    TORCH_CHECK(false, "unsupported pack dtype combination");
}

// ---------------------------------------------------------------------
// scale kernels (the unflatten+scale half of the bucket path)
// ---------------------------------------------------------------------
__global__ __launch_bounds__(kThreads) void scale_inplace_f32(
    float* __restrict__ p, long n, float s) {
  const long stride = (long)gridDim.x * kThreads * 4;
  for (long idx = ((long)blockIdx.x * kThreads + threadIdx.x) * 4;
       idx < n; idx += stride) {
    if (idx + 4 <= n) {
      float4 v = *reinterpret_cast<float4*>(p + idx);
      v.x *= s; v.y *= s; v.z *= s; v.w *= s;
      *reinterpret_cast<float4*>(p + idx) = v;
    } else {
      for (long i = idx; i < n; ++i) p[i] *= s;
    }
  }
}

__global__ __launch_bounds__(kThreads) void scale_inplace_bf16(
    __hip_bfloat16* __restrict__ p, long n, float s) {
  const long stride = (long)gridDim.x * kThreads * 8;
  for (long idx = ((long)blockIdx.x * kThreads + threadIdx.x) * 8;
       idx < n; idx += stride) {
    if (idx + 8 <= n) {
      __hip_bfloat16 v[8];
      *reinterpret_cast<int4*>(v) = *reinterpret_cast<int4*>(p + idx);
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        v[i] = __float2bfloat16(__bfloat162float(v[i]) * s);
      *reinterpret_cast<int4*>(p + idx) = *reinterpret_cast<int4*>(v);
    } else {
      for (long i = idx; i < n; ++i)
        p[i] = __float2bfloat16(__bfloat162float(p[i]) * s);
    }
  }
}

// dst_f32 = (float)src_bf16 * s  — fused upcast + 1/world scale
__global__ __launch_bounds__(kThreads) void scale_cast_kernel(
    float* __restrict__ dst, const __hip_bfloat16* __restrict__ src,
    long n, float s) {
  const long stride = (long)gridDim.x * kThreads * 4;
  for (long idx = ((long)blockIdx.x * kThreads + threadIdx.x) * 4;
       idx < n; idx += stride) {
    if (idx + 4 <= n) {
      __hip_bfloat16 v[4];
      *reinterpret_cast<int2*>(v) =
          *reinterpret_cast<const int2*>(src + idx);
      float4 o;
      o.x = __bfloat162float(v[0]) * s;
      o.y = __bfloat162float(v[1]) * s;
      o.z = __bfloat162float(v[2]) * s;
      o.w = __bfloat162float(v[3]) * s;
      *reinterpret_cast<float4*>(dst + idx) = o;
    } else {
      for (long i = idx; i < n; ++i)
        dst[i] = __bfloat162float(src[i]) * s;
    }
  }
}

static long grid_for(long n, int per_thread) {
  long blocks = (n + (long)kThreads * per_thread - 1) /
                ((long)kThreads * per_thread);
  // >= 2048 workgroups fills 256 CUs across 8 XCDs; cap to bound kernarg
  return std::min<long>(std::max<long>(blocks, 1), 32768);
}

void scale_inplace(torch::Tensor t, double s) {
  auto stream = at::hip::getCurrentHIPStream().stream();
  long n = t.numel();
  if (t.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(scale_inplace_f32, dim3(grid_for(n, 4)),
                       dim3(kThreads), 0, stream,
                       t.data_ptr<float>(), n, (float)s);
  } else if (t.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(scale_inplace_bf16, dim3(grid_for(n, 8)),
                       dim3(kThreads), 0, stream,
                       reinterpret_cast<__hip_bfloat16*>(t.data_ptr()), n,
                       (float)s);
  } else {
    TORCH_CHECK(false, "scale_inplace: unsupported dtype");
  }
  CHECK_HIP(hipGetLastError());
}

void scale_cast(torch::Tensor dst, torch::Tensor src, double s) {
  TORCH_CHECK(dst.scalar_type() == at::kFloat &&
              src.scalar_type() == at::kBFloat16,
              "scale_cast expects f32 dst, bf16 src");
  TORCH_CHECK(dst.numel() == src.numel());
  auto stream = at::hip::getCurrentHIPStream().stream();
  long n = dst.numel();
  hipLaunchKernelGGL(scale_cast_kernel, dim3(grid_for(n, 4)),
                     dim3(kThreads), 0, stream, dst.data_ptr<float>(),
                     reinterpret_cast<const __hip_bfloat16*>(src.data_ptr()),
                     n, (float)s);
  CHECK_HIP(hipGetLastError());
}

// ---------------------------------------------------------------------
// fused multi-tensor SGD (momentum)  — fp32 params/grads/momentum
// ---------------------------------------------------------------------
struct SgdMeta {
  float* p[kMaxSeg];
  const float* g[kMaxSeg];
  float* m[kMaxSeg];
  long numel[kMaxSeg];
  long start_block[kMaxSeg + 1];
  int n;
};

template <int UNROLL = 4>
__global__ __launch_bounds__(kThreads) void fused_sgd_kernel(
    SgdMeta meta, float lr, float momentum, float dampening,
    float weight_decay, int nesterov, int first_step) {
  int seg = 0;
  long b = blockIdx.x;
  while (seg + 1 < meta.n && b >= meta.start_block[seg + 1]) seg++;
  const long local_block = b - meta.start_block[seg];
  const long numel = meta.numel[seg];
  float* __restrict__ p = meta.p[seg];
  const float* __restrict__ g = meta.g[seg];
  float* __restrict__ m = meta.m[seg];

  const long block_elems = (long)kThreads * 4 * UNROLL;
  const long base = local_block * block_elems;
  #pragma unroll
  for (int u = 0; u < UNROLL; ++u) {
    const long idx = base + ((long)u * kThreads + threadIdx.x) * 4;
    if (idx + 4 <= numel) {
      float4 pv = *reinterpret_cast<float4*>(p + idx);
      float4 gv = *reinterpret_cast<const float4*>(g + idx);
      float4 mv = (momentum != 0.f)
          ? *reinterpret_cast<float4*>(m + idx) : float4{};
      float pe[4] = {pv.x, pv.y, pv.z, pv.w};
      float ge[4] = {gv.x, gv.y, gv.z, gv.w};
      float me[4] = {mv.x, mv.y, mv.z, mv.w};
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        float d = ge[i] + weight_decay * pe[i];
        if (momentum != 0.f) {
          me[i] = first_step ? d
                             : momentum * me[i] + (1.f - dampening) * d;
          d = nesterov ? d + momentum * me[i] : me[i];
        }
        pe[i] = pe[i] - lr * d;
      }
      *reinterpret_cast<float4*>(p + idx) =
          float4{pe[0], pe[1], pe[2], pe[3]};
      if (momentum != 0.f)
        *reinterpret_cast<float4*>(m + idx) =
            float4{me[0], me[1], me[2], me[3]};
    } else {
      for (long i = idx; i < numel && i < idx + 4; ++i) {
        float d = g[i] + weight_decay * p[i];
        if (momentum != 0.f) {
          float mi = first_step ? d
                                : momentum * m[i] + (1.f - dampening) * d;
          m[i] = mi;
          d = nesterov ? d + momentum * mi : mi;
        }
        p[i] -= lr * d;
      }
    }
  }
}

void fused_sgd(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> momenta, double lr,
               double momentum, double dampening, double weight_decay,
               bool nesterov, bool first_step) {
  if (params.empty()) return;
  auto stream = at::hip::getCurrentHIPStream().stream();
  constexpr int UNROLL = 4;
  const long block_elems = (long)kThreads * 4 * UNROLL;
  size_t i = 0;
  while (i < params.size()) {
    SgdMeta meta;
    int n = 0;
    long blocks = 0;
    while (i < params.size() && n < kMaxSeg) {
      meta.p[n] = params[i].data_ptr<float>();
      meta.g[n] = grads[i].data_ptr<float>();
      meta.m[n] = momenta.empty() ? nullptr
                                  : momenta[i].data_ptr<float>();
      meta.numel[n] = params[i].numel();
      meta.start_block[n] = blocks;
      blocks += (meta.numel[n] + block_elems - 1) / block_elems;
      ++n; ++i;
    }
    meta.start_block[n] = blocks;
    meta.n = n;
    hipLaunchKernelGGL((fused_sgd_kernel<UNROLL>), dim3(blocks),
                       dim3(kThreads), 0, stream, meta, (float)lr,
                       (float)momentum, (float)dampening,
                       (float)weight_decay, (int)nesterov,
                       (int)first_step);
    CHECK_HIP(hipGetLastError());
  }
}

// ---------------------------------------------------------------------
// fused multi-tensor Adam / AdamW — fp32 everything
// ---------------------------------------------------------------------
struct AdamMeta {
  float* p[kMaxSeg];
  const float* g[kMaxSeg];
  float* m[kMaxSeg];
  float* v[kMaxSeg];
  long numel[kMaxSeg];
  long start_block[kMaxSeg + 1];
  int n;
};

template <int UNROLL = 4>
__global__ __launch_bounds__(kThreads) void fused_adam_kernel(
    AdamMeta meta, float lr, float beta1, float beta2, float eps,
    float weight_decay, float bc1, float bc2, int adamw) {
  int seg = 0;
  long b = blockIdx.x;
  while (seg + 1 < meta.n && b >= meta.start_block[seg + 1]) seg++;
  const long local_block = b - meta.start_block[seg];
  const long numel = meta.numel[seg];
  float* __restrict__ p = meta.p[seg];
  const float* __restrict__ g = meta.g[seg];
  float* __restrict__ m = meta.m[seg];
  float* __restrict__ v = meta.v[seg];

  const float inv_bc1 = 1.f / bc1;
  const float inv_sqrt_bc2 = rsqrtf(bc2);

  const long block_elems = (long)kThreads * 4 * UNROLL;
  const long base = local_block * block_elems;
  #pragma unroll
  for (int u = 0; u < UNROLL; ++u) {
    const long idx = base + ((long)u * kThreads + threadIdx.x) * 4;
    if (idx + 4 <= numel) {
      float4 pv = *reinterpret_cast<float4*>(p + idx);
      float4 gv = *reinterpret_cast<const float4*>(g + idx);
      float4 mv = *reinterpret_cast<float4*>(m + idx);
      float4 vv = *reinterpret_cast<float4*>(v + idx);
      float pe[4] = {pv.x, pv.y, pv.z, pv.w};
      float ge[4] = {gv.x, gv.y, gv.z, gv.w};
      float me[4] = {mv.x, mv.y, mv.z, mv.w};
      float ve[4] = {vv.x, vv.y, vv.z, vv.w};
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        float gi = ge[i];
        if (adamw) pe[i] *= (1.f - lr * weight_decay);
        else gi += weight_decay * pe[i];
        me[i] = beta1 * me[i] + (1.f - beta1) * gi;
        ve[i] = beta2 * ve[i] + (1.f - beta2) * gi * gi;
        const float mhat = me[i] * inv_bc1;
        const float denom = sqrtf(ve[i]) * inv_sqrt_bc2 + eps;
        pe[i] -= lr * mhat / denom;
      }
      *reinterpret_cast<float4*>(p + idx) =
          float4{pe[0], pe[1], pe[2], pe[3]};
      *reinterpret_cast<float4*>(m + idx) =
          float4{me[0], me[1], me[2], me[3]};
      *reinterpret_cast<float4*>(v + idx) =
          float4{ve[0], ve[1], ve[2], ve[3]};
    } else {
      for (long i = idx; i < numel && i < idx + 4; ++i) {
        float gi = g[i];
        if (adamw) p[i] *= (1.f - lr * weight_decay);
        else gi += weight_decay * p[i];
        m[i] = beta1 * m[i] + (1.f - beta1) * gi;
        v[i] = beta2 * v[i] + (1.f - beta2) * gi * gi;
        p[i] -= lr * (m[i] * inv_bc1) /
                (sqrtf(v[i]) * inv_sqrt_bc2 + eps);
      }
    }
  }
}

void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps,
                double weight_decay, long step, bool adamw) {
  if (params.empty()) return;
  auto stream = at::hip::getCurrentHIPStream().stream();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  constexpr int UNROLL = 4;
  const long block_elems = (long)kThreads * 4 * UNROLL;
  size_t i = 0;
  while (i < params.size()) {
    AdamMeta meta;
    int n = 0;
    long blocks = 0;
    while (i < params.size() && n < kMaxSeg) {
      meta.p[n] = params[i].data_ptr<float>();
      meta.g[n] = grads[i].data_ptr<float>();
      meta.m[n] = exp_avgs[i].data_ptr<float>();
      meta.v[n] = exp_avg_sqs[i].data_ptr<float>();
      meta.numel[n] = params[i].numel();
      meta.start_block[n] = blocks;
      blocks += (meta.numel[n] + block_elems - 1) / block_elems;
      ++n; ++i;
    }
    meta.start_block[n] = blocks;
    meta.n = n;
    hipLaunchKernelGGL((fused_adam_kernel<UNROLL>), dim3(blocks),
                       dim3(kThreads), 0, stream, meta, (float)lr,
                       (float)beta1, (float)beta2, (float)eps,
                       (float)weight_decay, bc1, bc2, (int)adamw);
    CHECK_HIP(hipGetLastError());
  }
}

// ---------------------------------------------------------------------
// sharded fused Adam: bf16 model params + fp32 master copy (+ bf16 or
// f32 grads). The owner rank updates master state and writes back bf16
// params (which the sharded engine then broadcasts).
// ---------------------------------------------------------------------
struct ShardedAdamMeta {
  __hip_bfloat16* p16[kMaxSeg];
  float* master[kMaxSeg];
  const void* g[kMaxSeg];
  float* m[kMaxSeg];
  float* v[kMaxSeg];
  long numel[kMaxSeg];
  long start_block[kMaxSeg + 1];
  int n;
};

template <typename GradT, int UNROLL = 4>
__global__ __launch_bounds__(kThreads) void sharded_adam_kernel(
    ShardedAdamMeta meta, float lr, float beta1, float beta2, float eps,
    float weight_decay, float bc1, float bc2, int adamw) {
  int seg = 0;
  long b = blockIdx.x;
  while (seg + 1 < meta.n && b >= meta.start_block[seg + 1]) seg++;
  const long local_block = b - meta.start_block[seg];
  const long numel = meta.numel[seg];
  __hip_bfloat16* __restrict__ p16 = meta.p16[seg];
  float* __restrict__ master = meta.master[seg];
  const GradT* __restrict__ g =
      reinterpret_cast<const GradT*>(meta.g[seg]);
  float* __restrict__ m = meta.m[seg];
  float* __restrict__ v = meta.v[seg];

  const float inv_bc1 = 1.f / bc1;
  const float inv_sqrt_bc2 = rsqrtf(bc2);

  const long block_elems = (long)kThreads * 4 * UNROLL;
  const long base = local_block * block_elems;
  #pragma unroll
  for (int u = 0; u < UNROLL; ++u) {
    const long idx = base + ((long)u * kThreads + threadIdx.x) * 4;
    if (idx + 4 <= numel) {
      // vectorized: float4 state, 4-lane grads/params (a scalar v1 of
      // this loop was 3x off the bandwidth bound on GPT-2-XL)
      float4 mstv = *reinterpret_cast<float4*>(master + idx);
      float4 mv = *reinterpret_cast<float4*>(m + idx);
      float4 vv = *reinterpret_cast<float4*>(v + idx);
      GradT gv[4];
      if constexpr (sizeof(GradT) == 2)
        *reinterpret_cast<int2*>(gv) =
            *reinterpret_cast<const int2*>(g + idx);
      else
        *reinterpret_cast<int4*>(gv) =
            *reinterpret_cast<const int4*>(g + idx);
      float pe[4] = {mstv.x, mstv.y, mstv.z, mstv.w};
      float me[4] = {mv.x, mv.y, mv.z, mv.w};
      float ve[4] = {vv.x, vv.y, vv.z, vv.w};
      __hip_bfloat16 out16[4];
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        float gi = ToFloat<GradT>::conv(gv[i]);
        if (adamw) pe[i] *= (1.f - lr * weight_decay);
        else gi += weight_decay * pe[i];
        me[i] = beta1 * me[i] + (1.f - beta1) * gi;
        ve[i] = beta2 * ve[i] + (1.f - beta2) * gi * gi;
        pe[i] -= lr * (me[i] * inv_bc1) /
                 (sqrtf(ve[i]) * inv_sqrt_bc2 + eps);
        out16[i] = __float2bfloat16(pe[i]);
      }
      *reinterpret_cast<float4*>(master + idx) =
          float4{pe[0], pe[1], pe[2], pe[3]};
      *reinterpret_cast<float4*>(m + idx) =
          float4{me[0], me[1], me[2], me[3]};
      *reinterpret_cast<float4*>(v + idx) =
          float4{ve[0], ve[1], ve[2], ve[3]};
      *reinterpret_cast<int2*>(p16 + idx) =
          *reinterpret_cast<const int2*>(out16);
    } else {
      for (long i = idx; i < numel && i < idx + 4; ++i) {
        float gi = ToFloat<GradT>::conv(g[i]);
        float pi = master[i];
        if (adamw) pi *= (1.f - lr * weight_decay);
        else gi += weight_decay * pi;
        m[i] = beta1 * m[i] + (1.f - beta1) * gi;
        v[i] = beta2 * v[i] + (1.f - beta2) * gi * gi;
        pi -= lr * (m[i] * inv_bc1) /
              (sqrtf(v[i]) * inv_sqrt_bc2 + eps);
        master[i] = pi;
        p16[i] = __float2bfloat16(pi);
      }
    }
  }
}

void sharded_adam(std::vector<torch::Tensor> params_bf16,
                  std::vector<torch::Tensor> masters,
                  std::vector<torch::Tensor> grads,
                  std::vector<torch::Tensor> exp_avgs,
                  std::vector<torch::Tensor> exp_avg_sqs, double lr,
                  double beta1, double beta2, double eps,
                  double weight_decay, long step, bool adamw) {
  if (params_bf16.empty()) return;
  auto stream = at::hip::getCurrentHIPStream().stream();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  const bool grad_bf16 = grads[0].scalar_type() == at::kBFloat16;
  constexpr int UNROLL = 4;
  const long block_elems = (long)kThreads * 4 * UNROLL;
  size_t i = 0;
  while (i < params_bf16.size()) {
    ShardedAdamMeta meta;
    int n = 0;
    long blocks = 0;
    while (i < params_bf16.size() && n < kMaxSeg) {
      meta.p16[n] =
          reinterpret_cast<__hip_bfloat16*>(params_bf16[i].data_ptr());
      meta.master[n] = masters[i].data_ptr<float>();
      meta.g[n] = grads[i].data_ptr();
      meta.m[n] = exp_avgs[i].data_ptr<float>();
      meta.v[n] = exp_avg_sqs[i].data_ptr<float>();
      meta.numel[n] = params_bf16[i].numel();
      meta.start_block[n] = blocks;
      blocks += (meta.numel[n] + block_elems - 1) / block_elems;
      ++n; ++i;
    }
    meta.start_block[n] = blocks;
    meta.n = n;
    if (grad_bf16)
      hipLaunchKernelGGL((sharded_adam_kernel<__hip_bfloat16, UNROLL>),
                         dim3(blocks), dim3(kThreads), 0, stream, meta,
                         (float)lr, (float)beta1, (float)beta2,
                         (float)eps, (float)weight_decay, bc1, bc2,
                         (int)adamw);
    else
      hipLaunchKernelGGL((sharded_adam_kernel<float, UNROLL>),
                         dim3(blocks), dim3(kThreads), 0, stream, meta,
                         (float)lr, (float)beta1, (float)beta2,
                         (float)eps, (float)weight_decay, bc1, bc2,
                         (int)adamw);
    CHECK_HIP(hipGetLastError());
  }
}

// fused BN(+ReLU)(+residual) for channels_last — defined in fused_bn.hip
std::vector<torch::Tensor> fused_bn_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor running_mean, torch::Tensor running_var,
    c10::optional<torch::Tensor> res, bool relu, bool training,
    double momentum, double eps);
std::vector<torch::Tensor> fused_bn_bwd(
    torch::Tensor dy, torch::Tensor mask, torch::Tensor x,
    torch::Tensor mean, torch::Tensor invstd, torch::Tensor gamma,
    bool relu, bool has_res);

// flash attention (causal, hs=64) — defined in flash_attn.hip
std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q,
                                          torch::Tensor k,
                                          torch::Tensor v, double scale);
std::vector<torch::Tensor> flash_attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k,
    torch::Tensor v, torch::Tensor o, torch::Tensor lse, double scale);
std::vector<torch::Tensor> flash_attn_fwd_v3(torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v,
                                             double scale,
                                             bool use_permlane);
std::vector<torch::Tensor> flash_attn_bwd_v3(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k,
    torch::Tensor v, torch::Tensor o, torch::Tensor lse, double scale,
    bool use_permlane);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
torch::Tensor perm_probe(torch::Tensor M, torch::Tensor B,
                         long variant);
std::vector<torch::Tensor> perm_dump(torch::Tensor M, long variant);
torch::Tensor permlane_swap_probe();
torch::Tensor tr16_probe(long a, long b, long c);
torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor B);
std::vector<torch::Tensor> flash_attn_fwd_v4(torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v,
                                             double scale);

// fused residual-add + LayerNorm — defined in fused_ln.hip
std::vector<torch::Tensor> fused_ln_fwd(
    torch::Tensor a, c10::optional<torch::Tensor> b,
    torch::Tensor gamma, torch::Tensor beta, double eps);
std::vector<torch::Tensor> fused_ln_bwd(
    torch::Tensor dy, torch::Tensor x,
    c10::optional<torch::Tensor> dext, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor invstd);


// ===================================================================
// fused dGELU(tanh) + bias-gradient partials: one pass over [M, F]
// computes dz = dh * gelu'(z) AND per-slice column sums of dz, so the
// separate GeluBackward kernel + the full reduce_kernel re-read of dz
// collapse into one streaming pass + a tiny fold (the deterministic
// two-stage pattern from the fused-BN kernels).
// ===================================================================
namespace {

__global__ __launch_bounds__(256) void dgelu_bgrad_kernel(
    const __hip_bfloat16* __restrict__ dh,
    const __hip_bfloat16* __restrict__ z,
    __hip_bfloat16* __restrict__ dz, float* __restrict__ partial,
    long M, long F, long rows_per_slice) {
  // block: 256 threads x 2 cols each = 512 columns; grid.x = col
  // blocks, grid.y = row slices
  const long c0 = (long)blockIdx.x * 512 + threadIdx.x * 2;
  if (c0 >= F) return;
  const long r0 = (long)blockIdx.y * rows_per_slice;
  const long r1 = (r0 + rows_per_slice < M) ? r0 + rows_per_slice : M;
  const bool two = (c0 + 1) < F;
  float s0 = 0.f, s1 = 0.f;
  for (long r = r0; r < r1; ++r) {
    const long off = r * F + c0;
    const float zv0 = (float)z[off];
    const float g0 = (float)dh[off];
    // gelu'(x) for tanh approximation
    const float t0 = tanhf(0.7978845608028654f *
                           (zv0 + 0.044715f * zv0 * zv0 * zv0));
    const float d0 =
        0.5f * (1.0f + t0) +
        0.5f * zv0 * (1.0f - t0 * t0) * 0.7978845608028654f *
            (1.0f + 0.134145f * zv0 * zv0);
    const float o0 = g0 * d0;
    dz[off] = (__hip_bfloat16)o0;
    s0 += o0;
    if (two) {
      const float zv1 = (float)z[off + 1];
      const float g1 = (float)dh[off + 1];
      const float t1 = tanhf(0.7978845608028654f *
                             (zv1 + 0.044715f * zv1 * zv1 * zv1));
      const float d1 =
          0.5f * (1.0f + t1) +
          0.5f * zv1 * (1.0f - t1 * t1) * 0.7978845608028654f *
              (1.0f + 0.134145f * zv1 * zv1);
      const float o1 = g1 * d1;
      dz[off + 1] = (__hip_bfloat16)o1;
      s1 += o1;
    }
  }
  float* prow = partial + (long)blockIdx.y * F;
  prow[c0] = s0;
  if (two) prow[c0 + 1] = s1;
}

__global__ __launch_bounds__(256) void bgrad_fold_kernel(
    const float* __restrict__ partial, __hip_bfloat16* __restrict__ db,
    long F, long slices) {
  const long c = (long)blockIdx.x * 256 + threadIdx.x;
  if (c >= F) return;
  float s = 0.f;
  for (long i = 0; i < slices; ++i) s += partial[i * F + c];
  db[c] = (__hip_bfloat16)s;
}

}  // namespace

std::vector<torch::Tensor> fused_dgelu_bgrad(torch::Tensor dh,
                                             torch::Tensor z) {
  TORCH_CHECK(dh.is_cuda() && dh.scalar_type() == at::kBFloat16 &&
              dh.is_contiguous() && z.is_contiguous() &&
              dh.sizes() == z.sizes() && dh.dim() == 2);
  const long M = dh.size(0), F = dh.size(1);
  auto dz = torch::empty_like(dh);
  auto db = torch::empty({F}, dh.options());
  const long rows_per_slice = 512;
  const long slices = (M + rows_per_slice - 1) / rows_per_slice;
  auto partial = torch::empty({slices, F},
                              dh.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((F + 511) / 512, slices);
  hipLaunchKernelGGL(dgelu_bgrad_kernel, grid, dim3(256), 0, stream,
      reinterpret_cast<const __hip_bfloat16*>(dh.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(z.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(dz.data_ptr()),
      partial.data_ptr<float>(), M, F, rows_per_slice);
  hipLaunchKernelGGL(bgrad_fold_kernel, dim3((F + 255) / 256),
                     dim3(256), 0, stream, partial.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(db.data_ptr()),
                     F, slices);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "fused_dgelu_bgrad: ",
              hipGetErrorString(e));
  return {dz, db};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("multi_tensor_pack", &multi_tensor_pack,
        "multi-tensor flatten/cast into bucket slots");
  m.def("fused_bn_fwd", &fused_bn_fwd,
        "fused NHWC BN(+ReLU)(+residual) forward");
  m.def("fused_bn_bwd", &fused_bn_bwd,
        "fused NHWC BN(+ReLU)(+residual) backward");
  m.def("fused_ln_fwd", &fused_ln_fwd,
        "fused residual-add + LayerNorm forward");
  m.def("fused_ln_bwd", &fused_ln_bwd,
        "fused residual-add + LayerNorm backward");
  m.def("flash_attn_fwd", &flash_attn_fwd,
        "causal flash attention forward (hs=64, bf16, MFMA)");
  m.def("flash_attn_fwd_v3", &flash_attn_fwd_v3,
        "swapped-operand MFMA flash attention fwd (v3)");
  m.def("flash_attn_bwd", &flash_attn_bwd,
        "causal flash attention backward");
  m.def("flash_attn_bwd_v3", &flash_attn_bwd_v3,
        "swapped-operand MFMA flash attention bwd (v3)");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("perm_probe", &perm_probe,
        "C-layout -> A-fragment bpermute redistribution probe");
  m.def("perm_dump", &perm_dump, "redistribution element dump");
  m.def("flash_attn_fwd_v4", &flash_attn_fwd_v4,
        "32x32x16 MFMA flash attention fwd (v4)");
  m.def("mfma_probe32", &mfma_probe32,
        "32x32x16 bf16 MFMA layout probe");
  m.def("tr16_probe", &tr16_probe,
        "ds_read_b64_tr_b16 lane/element map probe");
  m.def("permlane_swap_probe", &permlane_swap_probe,
        "permlane{16,32}_swap lane-exchange semantics probe");
  m.def("fused_dgelu_bgrad", &fused_dgelu_bgrad,
        "dz = dh * gelu_tanh'(z) + column-sum bias grad, one pass");
  m.def("scale_inplace", &scale_inplace, "flat *= s");
  m.def("scale_cast", &scale_cast, "dst_f32 = src_bf16 * s");
  m.def("fused_sgd", &fused_sgd, "fused multi-tensor SGD(momentum)");
  m.def("fused_adam", &fused_adam, "fused multi-tensor Adam/AdamW");
  m.def("sharded_adam", &sharded_adam,
        "sharded Adam: bf16 param + fp32 master");
}
