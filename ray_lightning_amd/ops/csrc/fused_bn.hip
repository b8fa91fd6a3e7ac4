// Fused BatchNorm(+ReLU)(+residual add) for NHWC (channels_last) on
// CDNA4. Replaces the MIOpen BN kernel quintet + ATen add/clamp chain
// that dominates the ResNet-50 step (see
// profiles/r01_resnet50_1gpu_fixedfind.md: BN + elementwise = >50% of
// step time; the convs themselves are only ~1/3).
//
// Memory-bound design: the train forward is 2 passes over x (stats +
// apply) instead of eager's 4 reads + 3 writes (BN stats, BN norm,
// add, relu); the backward is 2 passes (reduce + dx) instead of 5.
//
// Per-channel reductions are TWO-STAGE: stage 1 writes per-workgroup
// partials [nblocks, C] (coalesced stores, no atomics — a v1 of this
// file used one fp32 atomicAdd per (block, channel) and the ~8k
// serialized RMWs per address made stats 50x slower than MIOpen's);
// stage 2 folds partials and emits the per-channel coefficients the
// elementwise pass needs, so the hot elementwise kernels do float4
// coefficient loads instead of 16 scalar loads per 8 elements.
// Fixed stage-1 grid => deterministic accumulation order.
//
// Layout convention: x is [R, C] row-major with C contiguous
// (R = N*H*W): exactly torch channels_last.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_HIP_BN(cmd)                                                  \
  do {                                                                     \
    hipError_t e = (cmd);                                                  \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));     \
  } while (0)

namespace {

constexpr int kT = 256;        // 4 wave64 per workgroup
constexpr int kMaxStage1 = 1024;  // stage-1 workgroups (reduction grid)

template <typename T> struct BnVec;
template <> struct BnVec<float> { static constexpr int V = 4; };
template <> struct BnVec<__hip_bfloat16> { static constexpr int V = 8; };

template <typename T>
__device__ __forceinline__ float bn_tof(T v);
template <>
__device__ __forceinline__ float bn_tof<float>(float v) { return v; }
template <>
__device__ __forceinline__ float bn_tof<__hip_bfloat16>(
    __hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
__device__ __forceinline__ T bn_fromf(float v);
template <>
__device__ __forceinline__ float bn_fromf<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 bn_fromf<__hip_bfloat16>(
    float v) { return __float2bfloat16(v); }

template <typename T, int V>
__device__ __forceinline__ void load_vec(T (&dst)[V], const T* src) {
  *reinterpret_cast<int4*>(dst) = *reinterpret_cast<const int4*>(src);
}
template <typename T, int V>
__device__ __forceinline__ void store_vec(T* dst, const T (&src)[V]) {
  *reinterpret_cast<int4*>(dst) = *reinterpret_cast<const int4*>(src);
}
template <int V>
__device__ __forceinline__ void load_coef(float (&dst)[V],
                                          const float* src) {
  #pragma unroll
  for (int i = 0; i < V; i += 4)
    *reinterpret_cast<float4*>(dst + i) =
        *reinterpret_cast<const float4*>(src + i);
}

// --------------------------------------------------------------------
// stage 1 stats: per-(block, channel) partial sum / sumsq
// partial layout: [2, nblocks, C] (sum plane then sumsq plane)
// --------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(kT) void bn_stats_kernel(
    const T* __restrict__ x, float* __restrict__ partial, long R,
    int C) {
  constexpr int V = BnVec<T>::V;
  const int tpr = C / V;             // threads per row (<= kT)
  const int rpb = kT / tpr;          // rows per block (power of 2)
  const int slot = threadIdx.x % tpr;
  const int row_in_block = threadIdx.x / tpr;
  const int c0 = slot * V;

  float acc[V], acc2[V];
  #pragma unroll
  for (int i = 0; i < V; ++i) { acc[i] = 0.f; acc2[i] = 0.f; }

  const long row_stride = (long)gridDim.x * rpb;
  for (long r = (long)blockIdx.x * rpb + row_in_block; r < R;
       r += row_stride) {
    T v[V];
    load_vec<T, V>(v, x + r * C + c0);
    #pragma unroll
    for (int i = 0; i < V; ++i) {
      const float f = bn_tof<T>(v[i]);
      acc[i] += f;
      acc2[i] += f * f;
    }
  }

  __shared__ float lds[kT * BnVec<T>::V];
  const long nbC = (long)gridDim.x * C;
  // reduce sum then sumsq across the rpb rows sharing a channel slot
  for (int pass = 0; pass < 2; ++pass) {
    #pragma unroll
    for (int i = 0; i < V; ++i)
      lds[threadIdx.x * V + i] = pass ? acc2[i] : acc[i];
    __syncthreads();
    for (int s = rpb / 2; s > 0; s >>= 1) {
      if (row_in_block < s) {
        #pragma unroll
        for (int i = 0; i < V; ++i)
          lds[threadIdx.x * V + i] +=
              lds[(threadIdx.x + s * tpr) * V + i];
      }
      __syncthreads();
    }
    if (row_in_block == 0) {
      #pragma unroll
      for (int i = 0; i < V; ++i)
        partial[pass * nbC + (long)blockIdx.x * C + c0 + i] =
            lds[threadIdx.x * V + i];
    }
    __syncthreads();
  }
}

// --------------------------------------------------------------------
// stage 2a: fold the [nblocks, C] partials down to [kB2, C] with a 2-D
// grid (channel-groups x b-slices) so even C=64 layers get enough
// workgroups to hide memory latency (a v3 used one block per 64
// channels: 1 block on the whole chip for layer1, 74us latency-bound).
// Deterministic: fixed slice boundaries, no atomics.
// stage 2b: tiny per-channel kernel folds the kB2 rows and computes
// the coefficients.
// --------------------------------------------------------------------
constexpr int kFoldC = 64;   // channels per fold block
constexpr int kB2 = 16;      // second-level partial rows

__global__ __launch_bounds__(kT) void bn_fold2_kernel(
    const float* __restrict__ partial, int nblocks,
    float* __restrict__ partial2, int C) {
  // grid: (ceil(C/kFoldC), kB2); threads (kFoldC x lanes)
  const int lanes = kT / kFoldC;
  const int c = blockIdx.x * kFoldC + threadIdx.x % kFoldC;
  const int lane = threadIdx.x / kFoldC;
  const bool active = c < C;
  const int b_begin = (int)(((long)blockIdx.y * nblocks) / kB2);
  const int b_end = (int)(((long)(blockIdx.y + 1) * nblocks) / kB2);
  const long nbC = (long)nblocks * C;
  float s = 0.f, s2 = 0.f;
  if (active)
    for (int b = b_begin + lane; b < b_end; b += lanes) {
      s += partial[(long)b * C + c];
      s2 += partial[nbC + (long)b * C + c];
    }
  __shared__ float lsum[kT], lsq[kT];
  lsum[threadIdx.x] = s;
  lsq[threadIdx.x] = s2;
  __syncthreads();
  for (int st = lanes / 2; st > 0; st >>= 1) {
    if (lane < st) {
      lsum[threadIdx.x] += lsum[threadIdx.x + st * kFoldC];
      lsq[threadIdx.x] += lsq[threadIdx.x + st * kFoldC];
    }
    __syncthreads();
  }
  if (lane != 0 || !active) return;
  const long b2C = (long)kB2 * C;
  partial2[(long)blockIdx.y * C + c] = lsum[threadIdx.x];
  partial2[b2C + (long)blockIdx.y * C + c] = lsq[threadIdx.x];
}

__global__ void bn_stats_coef_kernel(
    const float* __restrict__ partial2,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    float* __restrict__ scale_out, float* __restrict__ bias_out,
    long R, int C, float momentum, float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long b2C = (long)kB2 * C;
  float s = 0.f, s2 = 0.f;
  #pragma unroll
  for (int b = 0; b < kB2; ++b) {
    s += partial2[(long)b * C + c];
    s2 += partial2[b2C + (long)b * C + c];
  }
  const float m = s / (float)R;
  const float var = fmaxf(s2 / (float)R - m * m, 0.f);
  const float inv = rsqrtf(var + eps);
  mean_out[c] = m;
  invstd_out[c] = inv;
  const float sc = gamma[c] * inv;
  scale_out[c] = sc;
  bias_out[c] = beta[c] - m * sc;
  if (running_mean != nullptr) {
    // unbiased variance for running stats (torch semantics)
    const float var_unb = R > 1 ? var * (float)R / (float)(R - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.f - momentum) * running_var[c] +
                     momentum * var_unb;
  }
}

// eval mode: scale/bias straight from running stats
__global__ void bn_eval_coef_kernel(
    const float* __restrict__ running_mean,
    const float* __restrict__ running_var,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ scale_out, float* __restrict__ bias_out, int C,
    float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float inv = rsqrtf(running_var[c] + eps);
  const float sc = gamma[c] * inv;
  scale_out[c] = sc;
  bias_out[c] = beta[c] - running_mean[c] * sc;
}

// --------------------------------------------------------------------
// apply: y = [relu](scale*x + bias [+ res])
// --------------------------------------------------------------------
template <typename T, bool RELU, bool RES, bool WMASK>
__global__ __launch_bounds__(kT) void bn_apply_kernel(
    const T* __restrict__ x, T* __restrict__ y,
    const float* __restrict__ scale, const float* __restrict__ bias,
    const T* __restrict__ res, unsigned char* __restrict__ mask,
    long total, int C) {
  constexpr int V = BnVec<T>::V;
  const long stride = (long)gridDim.x * kT;
  const long nvec = total / V;
  for (long vi = (long)blockIdx.x * kT + threadIdx.x; vi < nvec;
       vi += stride) {
    const long idx = vi * V;
    const int c0 = (int)(idx % C);
    T xv[V];
    load_vec<T, V>(xv, x + idx);
    float sc[V], bi[V];
    load_coef<V>(sc, scale + c0);
    load_coef<V>(bi, bias + c0);
    T rv[V];
    if constexpr (RES) load_vec<T, V>(rv, res + idx);
    T ov[V];
    unsigned int m = 0;
    #pragma unroll
    for (int i = 0; i < V; ++i) {
      float o = sc[i] * bn_tof<T>(xv[i]) + bi[i];
      if constexpr (RES) o += bn_tof<T>(rv[i]);
      if constexpr (RELU) {
        if constexpr (WMASK) m |= (o > 0.f ? 1u : 0u) << i;
        o = fmaxf(o, 0.f);
      }
      ov[i] = bn_fromf<T>(o);
    }
    store_vec<T, V>(y + idx, ov);
    if constexpr (WMASK) mask[vi] = (unsigned char)m;
  }
}

// --------------------------------------------------------------------
// backward stage 1: per-(block, channel) partial sum(dy_eff) and
// sum(dy_eff * xhat); dy_eff = relu ? dy * (y > 0) : dy
// partial layout: [2, nblocks, C]
// --------------------------------------------------------------------
template <typename T, bool RELU>
__global__ __launch_bounds__(kT) void bn_bwd_reduce_kernel(
    const T* __restrict__ dy, const unsigned char* __restrict__ mask,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ partial,
    long R, int C) {
  constexpr int V = BnVec<T>::V;
  const int tpr = C / V;
  const int rpb = kT / tpr;
  const int slot = threadIdx.x % tpr;
  const int row_in_block = threadIdx.x / tpr;
  const int c0 = slot * V;

  float mu[V], is[V];
  load_coef<V>(mu, mean + c0);
  load_coef<V>(is, invstd + c0);

  float a0[V], a1[V];
  #pragma unroll
  for (int i = 0; i < V; ++i) { a0[i] = 0.f; a1[i] = 0.f; }

  const long row_stride = (long)gridDim.x * rpb;
  for (long r = (long)blockIdx.x * rpb + row_in_block; r < R;
       r += row_stride) {
    T dv[V], xv[V];
    load_vec<T, V>(dv, dy + r * C + c0);
    load_vec<T, V>(xv, x + r * C + c0);
    unsigned int m = 0xffu;
    if constexpr (RELU) m = mask[(r * C + c0) / V];
    #pragma unroll
    for (int i = 0; i < V; ++i) {
      float d = bn_tof<T>(dv[i]);
      if constexpr (RELU) d = (m >> i) & 1u ? d : 0.f;
      const float xh = (bn_tof<T>(xv[i]) - mu[i]) * is[i];
      a0[i] += d;
      a1[i] += d * xh;
    }
  }

  __shared__ float lds[kT * BnVec<T>::V];
  const long nbC = (long)gridDim.x * C;
  for (int pass = 0; pass < 2; ++pass) {
    #pragma unroll
    for (int i = 0; i < V; ++i)
      lds[threadIdx.x * V + i] = pass ? a1[i] : a0[i];
    __syncthreads();
    for (int s = rpb / 2; s > 0; s >>= 1) {
      if (row_in_block < s) {
        #pragma unroll
        for (int i = 0; i < V; ++i)
          lds[threadIdx.x * V + i] +=
              lds[(threadIdx.x + s * tpr) * V + i];
      }
      __syncthreads();
    }
    if (row_in_block == 0) {
      #pragma unroll
      for (int i = 0; i < V; ++i)
        partial[pass * nbC + (long)blockIdx.x * C + c0 + i] =
            lds[threadIdx.x * V + i];
    }
    __syncthreads();
  }
}

// --------------------------------------------------------------------
// backward stage 2: fold partials -> dgamma/dbeta and the three
// per-channel dx coefficients:
//   dx = P*dy_eff + Q*x + S
//   P = gamma*invstd
//   Q = -P * invstd * (dxhat_sum/R)
//   S = -P * (dsum/R) - Q * mean
// --------------------------------------------------------------------
__global__ void bn_bwd_coef_kernel(
    const float* __restrict__ partial2,
    const float* __restrict__ gamma, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ dgamma,
    float* __restrict__ dbeta, float* __restrict__ coefP,
    float* __restrict__ coefQ, float* __restrict__ coefS, long R,
    int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long b2C = (long)kB2 * C;
  float dsum = 0.f, dxhat = 0.f;
  #pragma unroll
  for (int b = 0; b < kB2; ++b) {
    dsum += partial2[(long)b * C + c];
    dxhat += partial2[b2C + (long)b * C + c];
  }
  dbeta[c] = dsum;
  dgamma[c] = dxhat;
  const float invR = 1.f / (float)R;
  const float P = gamma[c] * invstd[c];
  const float Q = -P * invstd[c] * dxhat * invR;
  coefP[c] = P;
  coefQ[c] = Q;
  coefS[c] = -P * dsum * invR - Q * mean[c];
}

// --------------------------------------------------------------------
// backward dx elementwise: dx = P*dy_eff + Q*x + S [, dres = dy_eff]
// --------------------------------------------------------------------
template <typename T, bool RELU, bool RES>
__global__ __launch_bounds__(kT) void bn_bwd_dx_kernel(
    const T* __restrict__ dy, const unsigned char* __restrict__ mask,
    const T* __restrict__ x, const float* __restrict__ coefP,
    const float* __restrict__ coefQ, const float* __restrict__ coefS,
    T* __restrict__ dx, T* __restrict__ dres, long total, int C) {
  constexpr int V = BnVec<T>::V;
  const long stride = (long)gridDim.x * kT;
  const long nvec = total / V;
  for (long vi = (long)blockIdx.x * kT + threadIdx.x; vi < nvec;
       vi += stride) {
    const long idx = vi * V;
    const int c0 = (int)(idx % C);
    T dv[V], xv[V];
    load_vec<T, V>(dv, dy + idx);
    load_vec<T, V>(xv, x + idx);
    unsigned int m = 0xffu;
    if constexpr (RELU) m = mask[vi];
    float P[V], Q[V], S[V];
    load_coef<V>(P, coefP + c0);
    load_coef<V>(Q, coefQ + c0);
    load_coef<V>(S, coefS + c0);
    T dxv[V], drv[V];
    #pragma unroll
    for (int i = 0; i < V; ++i) {
      float d = bn_tof<T>(dv[i]);
      if constexpr (RELU) d = (m >> i) & 1u ? d : 0.f;
      if constexpr (RES) drv[i] = bn_fromf<T>(d);
      dxv[i] = bn_fromf<T>(P[i] * d + Q[i] * bn_tof<T>(xv[i]) + S[i]);
    }
    store_vec<T, V>(dx + idx, dxv);
    if constexpr (RES) store_vec<T, V>(dres + idx, drv);
  }
}

int bn_grid_rows(long R, int rpb) {
  long blocks = (R + rpb - 1) / rpb;
  return (int)std::min<long>(std::max<long>(blocks, 1), kMaxStage1);
}

long bn_grid_elems(long nvec) {
  long blocks = (nvec + kT - 1) / kT;
  return std::min<long>(std::max<long>(blocks, 1), 16384);
}

template <typename T>
void bn_fwd_impl(const torch::Tensor& x, torch::Tensor& y,
                 const torch::Tensor& gamma, const torch::Tensor& beta,
                 torch::Tensor& running_mean, torch::Tensor& running_var,
                 torch::Tensor& mean, torch::Tensor& invstd,
                 const c10::optional<torch::Tensor>& res,
                 torch::Tensor& mask, bool relu,
                 bool training, double momentum, double eps, long R,
                 int C, hipStream_t stream) {
  auto opts = gamma.options().dtype(at::kFloat);
  auto scale = torch::empty({C}, opts);
  auto bias = torch::empty({C}, opts);
  if (training) {
    const int tpr = C / BnVec<T>::V;
    const int rpb = kT / tpr;
    const int nb = bn_grid_rows(R, rpb);
    auto partial = torch::empty({2, nb, C}, opts);
    auto partial2 = torch::empty({2, kB2, C}, opts);
    hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(nb), dim3(kT), 0,
                       stream,
                       reinterpret_cast<const T*>(x.data_ptr()),
                       partial.data_ptr<float>(), R, C);
    CHECK_HIP_BN(hipGetLastError());
    hipLaunchKernelGGL(bn_fold2_kernel,
                       dim3((C + kFoldC - 1) / kFoldC, kB2), dim3(kT), 0,
                       stream, partial.data_ptr<float>(), nb,
                       partial2.data_ptr<float>(), C);
    CHECK_HIP_BN(hipGetLastError());
    hipLaunchKernelGGL(bn_stats_coef_kernel, dim3((C + 255) / 256),
                       dim3(256), 0, stream, partial2.data_ptr<float>(),
                       gamma.data_ptr<float>(),
                       beta.data_ptr<float>(),
                       running_mean.defined()
                           ? running_mean.data_ptr<float>() : nullptr,
                       running_var.defined()
                           ? running_var.data_ptr<float>() : nullptr,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       scale.data_ptr<float>(), bias.data_ptr<float>(),
                       R, C, (float)momentum, (float)eps);
    CHECK_HIP_BN(hipGetLastError());
  } else {
    hipLaunchKernelGGL(bn_eval_coef_kernel, dim3((C + 255) / 256),
                       dim3(256), 0, stream,
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       scale.data_ptr<float>(), bias.data_ptr<float>(),
                       C, (float)eps);
    CHECK_HIP_BN(hipGetLastError());
  }
  const long total = R * (long)C;
  const long grid = bn_grid_elems(total / BnVec<T>::V);
  const T* resp = res.has_value()
      ? reinterpret_cast<const T*>(res->data_ptr()) : nullptr;
  unsigned char* mp = mask.defined()
      ? mask.data_ptr<unsigned char>() : nullptr;
  #define APPLY(RELU_, RES_, WM_)                                       \
    hipLaunchKernelGGL((bn_apply_kernel<T, RELU_, RES_, WM_>),          \
                       dim3(grid), dim3(kT), 0, stream,                 \
                       reinterpret_cast<const T*>(x.data_ptr()),        \
                       reinterpret_cast<T*>(y.data_ptr()),              \
                       scale.data_ptr<float>(), bias.data_ptr<float>(), \
                       resp, mp, total, C)
  const bool wm = relu && mp != nullptr;
  if (relu && resp) { if (wm) APPLY(true, true, true);
                      else APPLY(true, true, false); }
  else if (relu) { if (wm) APPLY(true, false, true);
                   else APPLY(true, false, false); }
  else if (resp) APPLY(false, true, false);
  else APPLY(false, false, false);
  #undef APPLY
  CHECK_HIP_BN(hipGetLastError());
}

template <typename T>
void bn_bwd_impl(const torch::Tensor& dy, const torch::Tensor& mask,
                 const torch::Tensor& x, const torch::Tensor& mean,
                 const torch::Tensor& invstd, const torch::Tensor& gamma,
                 torch::Tensor& dx, torch::Tensor& dgamma,
                 torch::Tensor& dbeta,
                 const c10::optional<torch::Tensor>& dres, bool relu,
                 long R, int C, hipStream_t stream) {
  const int tpr = C / BnVec<T>::V;
  const int rpb = kT / tpr;
  const int nb = bn_grid_rows(R, rpb);
  auto fopts = gamma.options().dtype(at::kFloat);
  auto partial = torch::empty({2, nb, C}, fopts);
  auto partial2 = torch::empty({2, kB2, C}, fopts);
  const unsigned char* mp = mask.defined()
      ? mask.data_ptr<unsigned char>() : nullptr;
  #define RED(RELU_)                                                     \
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, RELU_>), dim3(nb),       \
                       dim3(kT), 0, stream,                              \
                       reinterpret_cast<const T*>(dy.data_ptr()),        \
                       mp,                                               \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       mean.data_ptr<float>(), invstd.data_ptr<float>(), \
                       partial.data_ptr<float>(), R, C)
  if (relu) RED(true); else RED(false);
  #undef RED
  CHECK_HIP_BN(hipGetLastError());

  auto coefP = torch::empty({C}, fopts);
  auto coefQ = torch::empty({C}, fopts);
  auto coefS = torch::empty({C}, fopts);
  hipLaunchKernelGGL(bn_fold2_kernel,
                     dim3((C + kFoldC - 1) / kFoldC, kB2), dim3(kT), 0,
                     stream, partial.data_ptr<float>(), nb,
                     partial2.data_ptr<float>(), C);
  CHECK_HIP_BN(hipGetLastError());
  hipLaunchKernelGGL(bn_bwd_coef_kernel, dim3((C + 255) / 256),
                     dim3(256), 0, stream, partial2.data_ptr<float>(),
                     gamma.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), dgamma.data_ptr<float>(),
                     dbeta.data_ptr<float>(), coefP.data_ptr<float>(),
                     coefQ.data_ptr<float>(), coefS.data_ptr<float>(), R,
                     C);
  CHECK_HIP_BN(hipGetLastError());

  const long total = R * (long)C;
  const long grid = bn_grid_elems(total / BnVec<T>::V);
  T* dresp = dres.has_value()
      ? reinterpret_cast<T*>(dres->data_ptr()) : nullptr;
  #define DX(RELU_, RES_)                                                \
    hipLaunchKernelGGL((bn_bwd_dx_kernel<T, RELU_, RES_>), dim3(grid),   \
                       dim3(kT), 0, stream,                              \
                       reinterpret_cast<const T*>(dy.data_ptr()),        \
                       mp,                                               \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       coefP.data_ptr<float>(), coefQ.data_ptr<float>(), \
                       coefS.data_ptr<float>(),                          \
                       reinterpret_cast<T*>(dx.data_ptr()), dresp,       \
                       total, C)
  if (relu && dresp) DX(true, true);
  else if (relu) DX(true, false);
  else if (dresp) DX(false, true);
  else DX(false, false);
  #undef DX
  CHECK_HIP_BN(hipGetLastError());
}

}  // namespace

// x: [N, C, H, W] channels_last (viewed as [R, C]); gamma/beta fp32.
// Returns (y, save_mean, save_invstd).
std::vector<torch::Tensor> fused_bn_fwd(
    torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor running_mean, torch::Tensor running_var,
    c10::optional<torch::Tensor> res, bool relu, bool training,
    double momentum, double eps) {
  TORCH_CHECK(x.dim() == 4, "fused_bn: 4-D input expected");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "fused_bn: channels_last input required");
  const int C = (int)x.size(1);
  const long R = x.numel() / C;
  const int V = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  TORCH_CHECK(C % V == 0 && C / V <= kT,
              "fused_bn: C=", C, " unsupported for vec", V);
  if (res.has_value()) {
    TORCH_CHECK(res->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                res->scalar_type() == x.scalar_type() &&
                res->sizes() == x.sizes(), "fused_bn: bad residual");
  }
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto y = torch::empty_like(x);  // keeps channels_last layout
  auto fopts = gamma.options().dtype(at::kFloat);
  auto mean = torch::empty({C}, fopts);
  auto invstd = torch::empty({C}, fopts);
  // ReLU bitmask (1 byte per V elems) replaces re-reading y in
  // backward: ~25% less bwd traffic
  torch::Tensor mask;
  if (relu && training)
    mask = torch::empty({x.numel() / V},
                        x.options().dtype(at::kByte));
  if (x.scalar_type() == at::kBFloat16)
    bn_fwd_impl<__hip_bfloat16>(x, y, gamma, beta, running_mean,
                                running_var, mean, invstd, res, mask,
                                relu, training, momentum, eps, R, C,
                                stream);
  else if (x.scalar_type() == at::kFloat)
    bn_fwd_impl<float>(x, y, gamma, beta, running_mean, running_var,
                       mean, invstd, res, mask, relu, training,
                       momentum, eps, R, C, stream);
  else
    TORCH_CHECK(false, "fused_bn: dtype must be bf16 or f32");
  if (!mask.defined())
    mask = torch::empty({0}, x.options().dtype(at::kByte));
  return {y, mean, invstd, mask};
}

// Returns (dx, dgamma, dbeta[, dres]). mask: ReLU bitmask from fwd
// (empty tensor when relu=false).
std::vector<torch::Tensor> fused_bn_bwd(
    torch::Tensor dy, torch::Tensor mask, torch::Tensor x,
    torch::Tensor mean, torch::Tensor invstd, torch::Tensor gamma,
    bool relu, bool has_res) {
  const int C = (int)x.size(1);
  const long R = x.numel() / C;
  TORCH_CHECK(!relu || mask.numel() > 0,
              "fused_bn_bwd: relu path needs the fwd mask");
  dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto dx = torch::empty_like(x);
  auto fopts = gamma.options().dtype(at::kFloat);
  auto dgamma = torch::empty({C}, fopts);
  auto dbeta = torch::empty({C}, fopts);
  c10::optional<torch::Tensor> dres;
  if (has_res) dres = torch::empty_like(x);
  if (x.scalar_type() == at::kBFloat16)
    bn_bwd_impl<__hip_bfloat16>(dy, mask, x, mean, invstd, gamma, dx,
                                dgamma, dbeta, dres, relu, R, C, stream);
  else
    bn_bwd_impl<float>(dy, mask, x, mean, invstd, gamma, dx, dgamma,
                       dbeta, dres, relu, R, C, stream);
  std::vector<torch::Tensor> out = {dx, dgamma, dbeta};
  if (has_res) out.push_back(*dres);
  return out;
}
