// Fused residual-add + LayerNorm for [R, C] rows (transformer
// pre-norm blocks) on CDNA4.
//
//   forward:  x_new = a + b        (residual + sublayer output)
//             y     = (x_new - mean_row) * invstd_row * g + beta
//   one kernel: reads a,b once, writes x_new,y once — replaces the
//   eager add (2R+1W) + ATen LN (1R+1W) chain and its backward's
//   grad_input + two-stage PartGradGammaBeta kernels.
//
//   backward: block-per-row dx kernel (row reductions in LDS) +
//             column-reduction kernel for dgamma/dbeta with the same
//             two-stage [nblocks, C] -> fold pattern as fused_bn.hip.
//
// dx = invstd * (dyg - mean_row(dyg) - xhat * mean_row(dyg * xhat)),
//   dyg = dy * gamma;   d(a) = d(b) = dx + d_extern(x_new).
//
// Row layout: C contiguous, bf16/f32; C % 8 == 0 (bf16) / % 4 (f32),
// C <= 16384.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_HIP_LN(cmd)                                                  \
  do {                                                                     \
    hipError_t e = (cmd);                                                  \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));     \
  } while (0)

namespace {

constexpr int kT = 256;

template <typename T> struct LnVec;
template <> struct LnVec<float> { static constexpr int V = 4; };
template <> struct LnVec<__hip_bfloat16> { static constexpr int V = 8; };

template <typename T>
__device__ __forceinline__ float ln_tof(T v);
template <>
__device__ __forceinline__ float ln_tof<float>(float v) { return v; }
template <>
__device__ __forceinline__ float ln_tof<__hip_bfloat16>(
    __hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
__device__ __forceinline__ T ln_fromf(float v);
template <>
__device__ __forceinline__ float ln_fromf<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 ln_fromf<__hip_bfloat16>(
    float v) { return __float2bfloat16(v); }

__device__ __forceinline__ float block_sum(float v, float* lds) {
  lds[threadIdx.x] = v;
  __syncthreads();
  for (int s = kT / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) lds[threadIdx.x] += lds[threadIdx.x + s];
    __syncthreads();
  }
  const float out = lds[0];
  __syncthreads();
  return out;
}

// --------------------------------------------------------------------
// forward: one block per row; optional residual fusion (B==nullptr ->
// plain LN of A).
// --------------------------------------------------------------------
template <typename T, bool RES>
__global__ __launch_bounds__(kT) void ln_fwd_kernel(
    const T* __restrict__ a, const T* __restrict__ b,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    T* __restrict__ x_new, T* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    long R, int C, float eps) {
  constexpr int V = LnVec<T>::V;
  const int nvec = C / V;   // vector slots per row
  __shared__ float lds[kT];

  for (long r = blockIdx.x; r < R; r += gridDim.x) {
    const T* arow = a + r * C;
    const T* brow = RES ? b + r * C : nullptr;
    T* xrow = RES ? x_new + r * C : nullptr;
    // pass 1: x = a (+ b); accumulate sum/sumsq; stage x in registers
    // when the row fits (C <= kT*V covers GPT-2 sizes), else re-read.
    float s = 0.f, s2 = 0.f;
    const bool fits = nvec <= kT;
    // local storage for one vector slot per thread (fits case)
    T xloc[LnVec<T>::V];
    for (int vi = threadIdx.x; vi < nvec; vi += kT) {
      T av[V];
      *reinterpret_cast<int4*>(av) =
          *reinterpret_cast<const int4*>(arow + vi * V);
      if constexpr (RES) {
        T bv[V];
        *reinterpret_cast<int4*>(bv) =
            *reinterpret_cast<const int4*>(brow + vi * V);
        #pragma unroll
        for (int i = 0; i < V; ++i)
          av[i] = ln_fromf<T>(ln_tof<T>(av[i]) + ln_tof<T>(bv[i]));
        *reinterpret_cast<int4*>(xrow + vi * V) =
            *reinterpret_cast<const int4*>(av);
      }
      #pragma unroll
      for (int i = 0; i < V; ++i) {
        const float f = ln_tof<T>(av[i]);
        s += f;
        s2 += f * f;
      }
      if (fits) {
        #pragma unroll
        for (int i = 0; i < V; ++i) xloc[i] = av[i];
      }
    }
    s = block_sum(s, lds);
    s2 = block_sum(s2, lds);
    const float mean = s / (float)C;
    const float var = fmaxf(s2 / (float)C - mean * mean, 0.f);
    const float invstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean_out[r] = mean;
      invstd_out[r] = invstd;
    }
    // pass 2: normalize + affine
    for (int vi = threadIdx.x; vi < nvec; vi += kT) {
      T av[V];
      if (fits && vi == threadIdx.x) {
        #pragma unroll
        for (int i = 0; i < V; ++i) av[i] = xloc[i];
      } else {
        const T* src = RES ? xrow : arow;
        *reinterpret_cast<int4*>(av) =
            *reinterpret_cast<const int4*>(src + vi * V);
      }
      float gv[V], bv2[V];
      #pragma unroll
      for (int i = 0; i < V; i += 4) {
        *reinterpret_cast<float4*>(gv + i) =
            *reinterpret_cast<const float4*>(gamma + vi * V + i);
        *reinterpret_cast<float4*>(bv2 + i) =
            *reinterpret_cast<const float4*>(beta + vi * V + i);
      }
      T ov[V];
      #pragma unroll
      for (int i = 0; i < V; ++i)
        ov[i] = ln_fromf<T>(
            (ln_tof<T>(av[i]) - mean) * invstd * gv[i] + bv2[i]);
      *reinterpret_cast<int4*>(y + r * C + vi * V) =
          *reinterpret_cast<const int4*>(ov);
    }
  }
}

// --------------------------------------------------------------------
// backward dx: block per row. dyg = dy * gamma;
// dx = invstd * (dyg - mean(dyg) - xhat * mean(dyg * xhat))
// dout_extern (grad flowing into x_new from its other consumer) is
// added when present.
// --------------------------------------------------------------------
template <typename T, bool EXT>
__global__ __launch_bounds__(kT) void ln_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ dext, const float* __restrict__ gamma,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    T* __restrict__ dx, long R, int C) {
  constexpr int V = LnVec<T>::V;
  const int nvec = C / V;
  __shared__ float lds[kT];

  for (long r = blockIdx.x; r < R; r += gridDim.x) {
    const float mu = mean[r];
    const float is = invstd[r];
    float s1 = 0.f, s2 = 0.f;
    for (int vi = threadIdx.x; vi < nvec; vi += kT) {
      T dv[V], xv[V];
      *reinterpret_cast<int4*>(dv) =
          *reinterpret_cast<const int4*>(dy + r * C + vi * V);
      *reinterpret_cast<int4*>(xv) =
          *reinterpret_cast<const int4*>(x + r * C + vi * V);
      float gv[V];
      #pragma unroll
      for (int i = 0; i < V; i += 4)
        *reinterpret_cast<float4*>(gv + i) =
            *reinterpret_cast<const float4*>(gamma + vi * V + i);
      #pragma unroll
      for (int i = 0; i < V; ++i) {
        const float dyg = ln_tof<T>(dv[i]) * gv[i];
        const float xh = (ln_tof<T>(xv[i]) - mu) * is;
        s1 += dyg;
        s2 += dyg * xh;
      }
    }
    s1 = block_sum(s1, lds) / (float)C;
    s2 = block_sum(s2, lds) / (float)C;
    for (int vi = threadIdx.x; vi < nvec; vi += kT) {
      T dv[V], xv[V];
      *reinterpret_cast<int4*>(dv) =
          *reinterpret_cast<const int4*>(dy + r * C + vi * V);
      *reinterpret_cast<int4*>(xv) =
          *reinterpret_cast<const int4*>(x + r * C + vi * V);
      T ev[V];
      if constexpr (EXT)
        *reinterpret_cast<int4*>(ev) =
            *reinterpret_cast<const int4*>(dext + r * C + vi * V);
      float gv[V];
      #pragma unroll
      for (int i = 0; i < V; i += 4)
        *reinterpret_cast<float4*>(gv + i) =
            *reinterpret_cast<const float4*>(gamma + vi * V + i);
      T ov[V];
      #pragma unroll
      for (int i = 0; i < V; ++i) {
        const float dyg = ln_tof<T>(dv[i]) * gv[i];
        const float xh = (ln_tof<T>(xv[i]) - mu) * is;
        float g = is * (dyg - s1 - xh * s2);
        if constexpr (EXT) g += ln_tof<T>(ev[i]);
        ov[i] = ln_fromf<T>(g);
      }
      *reinterpret_cast<int4*>(dx + r * C + vi * V) =
          *reinterpret_cast<const int4*>(ov);
    }
  }
}

// --------------------------------------------------------------------
// backward dgamma/dbeta: stage 1 like fused_bn (rows x channels
// partials), then the bn_fold pattern. Reuses the column-reduction
// structure: thread layout (C/V slots x rows-per-block).
// --------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(kT) void ln_bwd_gb_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    float* __restrict__ partial, long R, int C) {
  constexpr int V = LnVec<T>::V;
  const int nvec = C / V;
  // Each block owns a channel-slot range and strides over rows.
  // Layout: threads split as (slots_per_block x rows_per_block).
  const int spb = min(nvec, kT);       // slots handled per block row
  const int rpb = kT / spb;            // rows in flight
  const int slot0 = (int)(blockIdx.x % ((nvec + spb - 1) / spb)) * spb;
  const int nblk_c = (nvec + spb - 1) / spb;
  const int brow = (int)(blockIdx.x / nblk_c);
  const int nblk_r = (int)(gridDim.x / nblk_c);
  const int slot = slot0 + (int)(threadIdx.x % spb);
  const int rlane = (int)(threadIdx.x / spb);

  float dg[V], db[V];
  #pragma unroll
  for (int i = 0; i < V; ++i) { dg[i] = 0.f; db[i] = 0.f; }

  if (slot < nvec) {
    for (long r = (long)brow * rpb + rlane; r < R;
         r += (long)nblk_r * rpb) {
      const float mu = mean[r];
      const float is = invstd[r];
      T dv[V], xv[V];
      *reinterpret_cast<int4*>(dv) =
          *reinterpret_cast<const int4*>(dy + r * C + slot * V);
      *reinterpret_cast<int4*>(xv) =
          *reinterpret_cast<const int4*>(x + r * C + slot * V);
      #pragma unroll
      for (int i = 0; i < V; ++i) {
        const float d = ln_tof<T>(dv[i]);
        const float xh = (ln_tof<T>(xv[i]) - mu) * is;
        db[i] += d;
        dg[i] += d * xh;
      }
    }
  }

  __shared__ float lds[kT * LnVec<T>::V];
  const long pC = (long)gridDim.x * C;  // plane stride (grid, C)
  for (int pass = 0; pass < 2; ++pass) {
    #pragma unroll
    for (int i = 0; i < V; ++i)
      lds[threadIdx.x * V + i] = pass ? db[i] : dg[i];
    __syncthreads();
    for (int s = rpb / 2; s > 0; s >>= 1) {
      if (rlane < s) {
        #pragma unroll
        for (int i = 0; i < V; ++i)
          lds[threadIdx.x * V + i] +=
              lds[(threadIdx.x + s * spb) * V + i];
      }
      __syncthreads();
    }
    if (rlane == 0 && slot < nvec) {
      #pragma unroll
      for (int i = 0; i < V; ++i)
        partial[pass * pC + (long)blockIdx.x * C + slot * V + i] =
            lds[threadIdx.x * V + i];
    }
    __syncthreads();
  }
}

// 2-D fold (channel-groups x b-slices) -> [kLnB2, C] second-level
// partials, then a tiny per-channel kernel — the same structure as
// fused_bn's fold (a 1-thread-per-channel fold measured 74us
// latency-bound on GPT-2-XL's 776 calls/step).
constexpr int kLnFoldC = 64;
constexpr int kLnB2 = 16;

__global__ __launch_bounds__(256) void ln_fold2_kernel(
    const float* __restrict__ partial, int nblocks,
    float* __restrict__ partial2, int C) {
  const int lanes = 256 / kLnFoldC;
  const int c = blockIdx.x * kLnFoldC + (int)(threadIdx.x % kLnFoldC);
  const int lane = (int)(threadIdx.x / kLnFoldC);
  const bool active = c < C;
  const int b_begin = (int)(((long)blockIdx.y * nblocks) / kLnB2);
  const int b_end = (int)(((long)(blockIdx.y + 1) * nblocks) / kLnB2);
  const long pC = (long)nblocks * C;
  float dg = 0.f, db = 0.f;
  if (active)
    for (int b = b_begin + lane; b < b_end; b += lanes) {
      dg += partial[(long)b * C + c];
      db += partial[pC + (long)b * C + c];
    }
  __shared__ float l1[256], l2[256];
  l1[threadIdx.x] = dg;
  l2[threadIdx.x] = db;
  __syncthreads();
  for (int st = lanes / 2; st > 0; st >>= 1) {
    if (lane < st) {
      l1[threadIdx.x] += l1[threadIdx.x + st * kLnFoldC];
      l2[threadIdx.x] += l2[threadIdx.x + st * kLnFoldC];
    }
    __syncthreads();
  }
  if (lane != 0 || !active) return;
  const long p2C = (long)kLnB2 * C;
  partial2[(long)blockIdx.y * C + c] = l1[threadIdx.x];
  partial2[p2C + (long)blockIdx.y * C + c] = l2[threadIdx.x];
}

__global__ void ln_gb_fold_kernel(
    const float* __restrict__ partial2, float* __restrict__ dgamma,
    float* __restrict__ dbeta, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long p2C = (long)kLnB2 * C;
  float dg = 0.f, db = 0.f;
  #pragma unroll
  for (int b = 0; b < kLnB2; ++b) {
    dg += partial2[(long)b * C + c];
    db += partial2[p2C + (long)b * C + c];
  }
  dgamma[c] = dg;
  dbeta[c] = db;
}

long ln_rows_grid(long R) {
  return std::min<long>(std::max<long>(R, 1), 8192);
}

}  // namespace

// a: [R, C]; b: optional residual. Returns (y, x_new, mean, invstd)
// (x_new aliases a's values when b is absent — no copy made).
std::vector<torch::Tensor> fused_ln_fwd(
    torch::Tensor a, c10::optional<torch::Tensor> b,
    torch::Tensor gamma, torch::Tensor beta, double eps) {
  TORCH_CHECK(a.is_contiguous(), "fused_ln: contiguous input required");
  const int C = (int)a.size(-1);
  const long R = a.numel() / C;
  const int V = a.scalar_type() == at::kBFloat16 ? 8 : 4;
  TORCH_CHECK(C % V == 0, "fused_ln: C % ", V, " != 0");
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto y = torch::empty_like(a);
  auto fopts = gamma.options().dtype(at::kFloat);
  auto mean = torch::empty({R}, fopts);
  auto invstd = torch::empty({R}, fopts);
  torch::Tensor x_new =
      b.has_value() ? torch::empty_like(a) : a;
  const long grid = ln_rows_grid(R);

  #define LAUNCH_FWD(T, RES)                                              \
    hipLaunchKernelGGL((ln_fwd_kernel<T, RES>), dim3(grid), dim3(kT), 0, \
        stream, reinterpret_cast<const T*>(a.data_ptr()),                \
        RES ? reinterpret_cast<const T*>(b->data_ptr()) : nullptr,       \
        gamma.data_ptr<float>(), beta.data_ptr<float>(),                 \
        RES ? reinterpret_cast<T*>(x_new.data_ptr()) : nullptr,          \
        reinterpret_cast<T*>(y.data_ptr()), mean.data_ptr<float>(),      \
        invstd.data_ptr<float>(), R, C, (float)eps)
  if (a.scalar_type() == at::kBFloat16) {
    if (b.has_value()) LAUNCH_FWD(__hip_bfloat16, true);
    else LAUNCH_FWD(__hip_bfloat16, false);
  } else if (a.scalar_type() == at::kFloat) {
    if (b.has_value()) LAUNCH_FWD(float, true);
    else LAUNCH_FWD(float, false);
  } else {
    TORCH_CHECK(false, "fused_ln: dtype must be bf16 or f32");
  }
  #undef LAUNCH_FWD
  CHECK_HIP_LN(hipGetLastError());
  return {y, x_new, mean, invstd};
}

// Returns (dx, dgamma, dbeta). dext: optional extra grad into x_new.
std::vector<torch::Tensor> fused_ln_bwd(
    torch::Tensor dy, torch::Tensor x,
    c10::optional<torch::Tensor> dext, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor invstd) {
  const int C = (int)x.size(-1);
  const long R = x.numel() / C;
  dy = dy.contiguous();
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto dx = torch::empty_like(x);
  auto fopts = gamma.options().dtype(at::kFloat);
  auto dgamma = torch::empty({C}, fopts);
  auto dbeta = torch::empty({C}, fopts);
  const long grid = ln_rows_grid(R);

  #define LAUNCH_DX(T, EXT)                                               \
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, EXT>), dim3(grid), dim3(kT), \
        0, stream, reinterpret_cast<const T*>(dy.data_ptr()),            \
        reinterpret_cast<const T*>(x.data_ptr()),                        \
        EXT ? reinterpret_cast<const T*>(dext->data_ptr()) : nullptr,    \
        gamma.data_ptr<float>(), mean.data_ptr<float>(),                 \
        invstd.data_ptr<float>(), reinterpret_cast<T*>(dx.data_ptr()),   \
        R, C)
  const bool bf16 = x.scalar_type() == at::kBFloat16;
  if (bf16) {
    if (dext.has_value()) LAUNCH_DX(__hip_bfloat16, true);
    else LAUNCH_DX(__hip_bfloat16, false);
  } else {
    if (dext.has_value()) LAUNCH_DX(float, true);
    else LAUNCH_DX(float, false);
  }
  #undef LAUNCH_DX
  CHECK_HIP_LN(hipGetLastError());

  // dgamma/dbeta: column reduction
  const int V = bf16 ? 8 : 4;
  const int nvec = C / V;
  const int spb = std::min(nvec, kT);
  const int rpb = kT / spb;
  const int nblk_c = (nvec + spb - 1) / spb;
  int nblk_r = (int)std::min<long>((R + rpb - 1) / rpb, 256);
  const int nb = nblk_c * nblk_r;
  auto partial = torch::zeros({2, nb, C}, fopts);
  auto partial2 = torch::empty({2, kLnB2, C}, fopts);
  #define LAUNCH_GB(T)                                                    \
    hipLaunchKernelGGL((ln_bwd_gb_kernel<T>), dim3(nb), dim3(kT), 0,     \
        stream, reinterpret_cast<const T*>(dy.data_ptr()),               \
        reinterpret_cast<const T*>(x.data_ptr()),                        \
        mean.data_ptr<float>(), invstd.data_ptr<float>(),                \
        partial.data_ptr<float>(), R, C)
  if (bf16) LAUNCH_GB(__hip_bfloat16); else LAUNCH_GB(float);
  #undef LAUNCH_GB
  CHECK_HIP_LN(hipGetLastError());
  hipLaunchKernelGGL(ln_fold2_kernel,
                     dim3((C + kLnFoldC - 1) / kLnFoldC, kLnB2),
                     dim3(256), 0, stream, partial.data_ptr<float>(),
                     nb, partial2.data_ptr<float>(), C);
  CHECK_HIP_LN(hipGetLastError());
  hipLaunchKernelGGL(ln_gb_fold_kernel, dim3((C + 255) / 256), dim3(256),
                     0, stream, partial2.data_ptr<float>(),
                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                     C);
  CHECK_HIP_LN(hipGetLastError());
  return {dx, dgamma, dbeta};
}
