// Transformer kernels for MI355X (gfx950): fused LayerNorm, bias+GELU and
// masked row-softmax, forward + backward, bf16/fp32 IO with fp32 math.
//
// These cover the fused elementwise/normalization work of the LineVul
// RoBERTa encoder and the CodeT5 stack; the plain
// GEMMs (QKV/FFN projections) go to rocBLAS/hipBLASLt, the weight-grad
// GEMMs to csrc/wgrad.hip when hipBLASLt's picks are pathological.
//
// Geometry notes: hidden D = 768 (12 elems/lane on a 64-wide wave) or 3072;
// softmax rows are seq-length 512 (8/lane). One wave per row everywhere,
// 4 rows per 256-thread block; fp32 statistics.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <algorithm>
using std::min;

#define WAVE 64

template <typename T> __device__ __forceinline__ float tf(T v);
template <> __device__ __forceinline__ float tf<float>(float v) { return v; }
template <> __device__ __forceinline__ float tf<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T> __device__ __forceinline__ T ff(float v);
template <> __device__ __forceinline__ float ff<float>(float v) { return v; }
template <> __device__ __forceinline__ __hip_bfloat16 ff<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off));
  return __shfl(v, 0);
}

// ---------------------------------------------------------------------------
// LayerNorm: y = (x - mean) * rstd * gamma + beta  (per row of D)
// ---------------------------------------------------------------------------

template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     T* __restrict__ y, float* __restrict__ mean,
                                     float* __restrict__ rstd, long N, int D,
                                     float eps) {
  const long row = (long)(blockIdx.x * (blockDim.x / WAVE)) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= N) return;
  const T* xr = x + row * D;
  float s = 0.f, s2 = 0.f;
  for (int d = lane; d < D; d += WAVE) {
    const float v = tf(xr[d]);
    s += v;
    s2 += v * v;
  }
  s = wave_sum(s);
  s2 = wave_sum(s2);
  const float mu = s / D;
  const float var = s2 / D - mu * mu;
  const float rs = rsqrtf(var + eps);
  if (lane == 0) {
    mean[row] = mu;
    rstd[row] = rs;
  }
  T* yr = y + row * D;
  for (int d = lane; d < D; d += WAVE)
    yr[d] = ff<T>((tf(xr[d]) - mu) * rs * gamma[d] + beta[d]);
}

// dx = rstd * (dyg - mean_d(dyg) - xhat * mean_d(dyg * xhat)), dyg = dy*gamma
template <typename T>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dx, long N, int D) {
  const long row = (long)(blockIdx.x * (blockDim.x / WAVE)) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= N) return;
  const T* dyr = dy + row * D;
  const T* xr = x + row * D;
  const float mu = mean[row], rs = rstd[row];
  float c1 = 0.f, c2 = 0.f;
  for (int d = lane; d < D; d += WAVE) {
    const float xh = (tf(xr[d]) - mu) * rs;
    const float g = tf(dyr[d]) * gamma[d];
    c1 += g;
    c2 += g * xh;
  }
  c1 = wave_sum(c1) / D;
  c2 = wave_sum(c2) / D;
  T* dxr = dx + row * D;
  for (int d = lane; d < D; d += WAVE) {
    const float xh = (tf(xr[d]) - mu) * rs;
    const float g = tf(dyr[d]) * gamma[d];
    dxr[d] = ff<T>(rs * (g - c1 - xh * c2));
  }
}

// per-column reductions: dgamma[d] = sum_rows dy*xhat, dbeta[d] = sum_rows dy
template <typename T>
__global__ void layernorm_wgrad_kernel(const T* __restrict__ dy,
                                       const T* __restrict__ x,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ rstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta, long N, int D,
                                       int rows_per_block) {
  const int d = blockIdx.x * blockDim.x + threadIdx.x;
  if (d >= D) return;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min((long)(r0 + rows_per_block), N);
  float dg = 0.f, db = 0.f;
  for (long r = r0; r < r1; ++r) {
    const float g = tf(dy[r * D + d]);
    dg += g * (tf(x[r * D + d]) - mean[r]) * rstd[r];
    db += g;
  }
  atomicAdd(dgamma + d, dg);
  atomicAdd(dbeta + d, db);
}

// ---------------------------------------------------------------------------
// bias + GELU (erf form, matching torch F.gelu / HF "gelu")
// ---------------------------------------------------------------------------

__device__ __forceinline__ float gelu_f(float v) {
  return 0.5f * v * (1.0f + erff(v * 0.70710678118654752f));
}
__device__ __forceinline__ float gelu_grad_f(float v) {
  const float cdf = 0.5f * (1.0f + erff(v * 0.70710678118654752f));
  const float pdf = 0.3989422804014327f * __expf(-0.5f * v * v);
  return cdf + v * pdf;
}

template <typename T>
__global__ void bias_gelu_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ y, long total, int D) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int d = (int)(i % D);
    y[i] = ff<T>(gelu_f(tf(x[i]) + bias[d]));
  }
}

template <typename T>
__global__ void bias_gelu_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ bias,
                                     T* __restrict__ dx, long total, int D) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int d = (int)(i % D);
    dx[i] = ff<T>(tf(dy[i]) * gelu_grad_f(tf(x[i]) + bias[d]));
  }
}

// ---------------------------------------------------------------------------
// Masked scaled row-softmax over attention scores.
// S (R, L) where R = B*H*L rows; key positions >= valid[b] get -inf.
// P = softmax(S * scale + mask). One wave per row.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void softmax_mask_fwd_kernel(const T* __restrict__ S,
                                        const int* __restrict__ valid,
                                        T* __restrict__ P, long R, int L,
                                        int rows_per_batch, float scale) {
  const long row = (long)(blockIdx.x * (blockDim.x / WAVE)) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= R) return;
  const int vl = valid ? valid[row / rows_per_batch] : L;
  const T* sr = S + row * L;
  float m = -3.4e38f;
  for (int j = lane; j < vl; j += WAVE) m = fmaxf(m, tf(sr[j]) * scale);
  m = wave_max(m);
  float sum = 0.f;
  for (int j = lane; j < vl; j += WAVE) sum += __expf(tf(sr[j]) * scale - m);
  sum = wave_sum(sum);
  const float inv = (sum > 0.f) ? 1.0f / sum : 0.f;
  T* pr = P + row * L;
  for (int j = lane; j < L; j += WAVE)
    pr[j] = ff<T>(j < vl ? __expf(tf(sr[j]) * scale - m) * inv : 0.f);
}

// dS = scale * P * (dP - sum_j(dP * P))
template <typename T>
__global__ void softmax_mask_bwd_kernel(const T* __restrict__ dP,
                                        const T* __restrict__ P,
                                        T* __restrict__ dS, long R, int L,
                                        float scale) {
  const long row = (long)(blockIdx.x * (blockDim.x / WAVE)) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= R) return;
  const T* dpr = dP + row * L;
  const T* pr = P + row * L;
  float dot = 0.f;
  for (int j = lane; j < L; j += WAVE) dot += tf(dpr[j]) * tf(pr[j]);
  dot = wave_sum(dot);
  T* dsr = dS + row * L;
  for (int j = lane; j < L; j += WAVE)
    dsr[j] = ff<T>(scale * tf(pr[j]) * (tf(dpr[j]) - dot));
}

// ---------------------------------------------------------------------------
// Launchers
// ---------------------------------------------------------------------------

#define ROWS_PER_BLOCK 4

template <typename T>
void launch_layernorm_fwd(const T* x, const float* gamma, const float* beta,
                          T* y, float* mean, float* rstd, long N, int D,
                          float eps, hipStream_t stream) {
  const int grid = (int)((N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  if (grid)
    hipLaunchKernelGGL(layernorm_fwd_kernel<T>, dim3(grid),
                       dim3(WAVE * ROWS_PER_BLOCK), 0, stream, x, gamma, beta, y,
                       mean, rstd, N, D, eps);
}

template <typename T>
void launch_layernorm_bwd(const T* dy, const T* x, const float* gamma,
                          const float* mean, const float* rstd, T* dx, long N,
                          int D, hipStream_t stream) {
  const int grid = (int)((N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  if (grid)
    hipLaunchKernelGGL(layernorm_bwd_kernel<T>, dim3(grid),
                       dim3(WAVE * ROWS_PER_BLOCK), 0, stream, dy, x, gamma,
                       mean, rstd, dx, N, D);
}

template <typename T>
void launch_layernorm_wgrad(const T* dy, const T* x, const float* mean,
                            const float* rstd, float* dgamma, float* dbeta,
                            long N, int D, hipStream_t stream) {
  const int block = 256;
  const int colb = (D + block - 1) / block;
  const int rows_per_block = 64;
  const int rowb = (int)((N + rows_per_block - 1) / rows_per_block);
  if (colb && rowb)
    hipLaunchKernelGGL(layernorm_wgrad_kernel<T>, dim3(colb, rowb), dim3(block),
                       0, stream, dy, x, mean, rstd, dgamma, dbeta, N, D,
                       rows_per_block);
}

template <typename T>
void launch_bias_gelu_fwd(const T* x, const float* bias, T* y, long total,
                          int D, hipStream_t stream) {
  const int block = 256;
  const int grid = (int)min((total + block - 1) / block, (long)4096);
  if (grid)
    hipLaunchKernelGGL(bias_gelu_fwd_kernel<T>, dim3(grid), dim3(block), 0,
                       stream, x, bias, y, total, D);
}

template <typename T>
void launch_bias_gelu_bwd(const T* dy, const T* x, const float* bias, T* dx,
                          long total, int D, hipStream_t stream) {
  const int block = 256;
  const int grid = (int)min((total + block - 1) / block, (long)4096);
  if (grid)
    hipLaunchKernelGGL(bias_gelu_bwd_kernel<T>, dim3(grid), dim3(block), 0,
                       stream, dy, x, bias, dx, total, D);
}

template <typename T>
void launch_softmax_mask_fwd(const T* S, const int* valid, T* P, long R, int L,
                             int rows_per_batch, float scale,
                             hipStream_t stream) {
  const int grid = (int)((R + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  if (grid)
    hipLaunchKernelGGL(softmax_mask_fwd_kernel<T>, dim3(grid),
                       dim3(WAVE * ROWS_PER_BLOCK), 0, stream, S, valid, P, R, L,
                       rows_per_batch, scale);
}

template <typename T>
void launch_softmax_mask_bwd(const T* dP, const T* P, T* dS, long R, int L,
                             float scale, hipStream_t stream) {
  const int grid = (int)((R + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  if (grid)
    hipLaunchKernelGGL(softmax_mask_bwd_kernel<T>, dim3(grid),
                       dim3(WAVE * ROWS_PER_BLOCK), 0, stream, dP, P, dS, R, L,
                       scale);
}

#define INSTANTIATE_TK(T)                                                    \
  template void launch_layernorm_fwd<T>(const T*, const float*, const float*, \
                                        T*, float*, float*, long, int, float, \
                                        hipStream_t);                         \
  template void launch_layernorm_bwd<T>(const T*, const T*, const float*,     \
                                        const float*, const float*, T*, long, \
                                        int, hipStream_t);                    \
  template void launch_layernorm_wgrad<T>(const T*, const T*, const float*,   \
                                          const float*, float*, float*, long, \
                                          int, hipStream_t);                  \
  template void launch_bias_gelu_fwd<T>(const T*, const float*, T*, long,     \
                                        int, hipStream_t);                    \
  template void launch_bias_gelu_bwd<T>(const T*, const T*, const float*, T*, \
                                        long, int, hipStream_t);              \
  template void launch_softmax_mask_fwd<T>(const T*, const int*, T*, long,    \
                                           int, int, float, hipStream_t);     \
  template void launch_softmax_mask_bwd<T>(const T*, const T*, T*, long, int, \
                                           float, hipStream_t);

INSTANTIATE_TK(float)
INSTANTIATE_TK(__hip_bfloat16)
