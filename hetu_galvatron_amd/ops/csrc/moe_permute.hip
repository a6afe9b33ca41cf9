// MoE token permute / unpermute kernels for gfx950.
//
// Replaces the reference's Triton permute/unpermute/sort kernels
// (galvatron/core/runtime/moe/fused_kernels.py:199-991): expert-sorted
// gather, probability-weighted top-k merge, and their backwards.
// Memory-bound row movers: 16 B/lane packets, grid-stride.
#include "common.h"

namespace {

// y[i] = x[rows[i]]
template <typename T>
__global__ void permute_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const long* __restrict__ rows, long m, int h) {
  const long packs = (long)m * (h / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < packs;
       idx += (long)gridDim.x * blockDim.x) {
    const long i = idx / (h / 8);
    const int c = (int)(idx - i * (h / 8)) * 8;
    float v[8];
    VecIO<T>::load(v, x + rows[i] * h + c);
    VecIO<T>::store(y + i * h + c, v);
  }
}

// dx[rows[i]] += dy[i]  (rows repeat across top-k: fp32 accumulator)
template <typename T>
__global__ void permute_bwd_kernel(const T* __restrict__ dy,
                                   float* __restrict__ acc,
                                   const long* __restrict__ rows, long m,
                                   int h) {
  const long packs = (long)m * (h / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < packs;
       idx += (long)gridDim.x * blockDim.x) {
    const long i = idx / (h / 8);
    const int c = (int)(idx - i * (h / 8)) * 8;
    float v[8];
    VecIO<T>::load(v, dy + i * h + c);
    float* a = acc + rows[i] * h + c;
#pragma unroll
    for (int j = 0; j < 8; ++j) unsafeAtomicAdd(a + j, v[j]);
  }
}

// out[t] = sum_j probs[i_j] * back[i_j],  i_j = inv_order[t*k + j]
template <typename T>
__global__ void unpermute_kernel(const T* __restrict__ back,
                                 const float* __restrict__ probs,
                                 const long* __restrict__ inv_order,
                                 T* __restrict__ out, long n, int k, int h) {
  const long packs = n * (h / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < packs;
       idx += (long)gridDim.x * blockDim.x) {
    const long t = idx / (h / 8);
    const int c = (int)(idx - t * (h / 8)) * 8;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int j = 0; j < k; ++j) {
      const long i = inv_order[t * k + j];
      const float p = probs[i];
      float v[8];
      VecIO<T>::load(v, back + i * h + c);
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[q] += p * v[q];
    }
    VecIO<T>::store(out + t * h + c, acc);
  }
}

// dback[i] = probs[i] * dout[t_i]; dprobs[i] = <dout[t_i], back[i]>
template <typename T>
__global__ void unpermute_bwd_kernel(const T* __restrict__ dout,
                                     const T* __restrict__ back,
                                     const float* __restrict__ probs,
                                     const long* __restrict__ t_of,  // [m]
                                     T* __restrict__ dback,
                                     float* __restrict__ dprobs_part,
                                     long m, int h) {
  const long packs = (long)m * (h / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < packs;
       idx += (long)gridDim.x * blockDim.x) {
    const long i = idx / (h / 8);
    const int c = (int)(idx - i * (h / 8)) * 8;
    const float p = probs[i];
    float g[8], b[8], o[8];
    VecIO<T>::load(g, dout + t_of[i] * h + c);
    VecIO<T>::load(b, back + i * h + c);
    float dot = 0.f;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      o[q] = p * g[q];
      dot += g[q] * b[q];
    }
    VecIO<T>::store(dback + i * h + c, o);
    unsafeAtomicAdd(dprobs_part + i, dot);
  }
}

}  // namespace

#define DEF_LAUNCH(T)                                                         \
  void moe_permute_launch_##T(const T* x, T* y, const long* rows, long m,     \
                              int h, hipStream_t s) {                         \
    long packs = m * (h / 8);                                                 \
    hipLaunchKernelGGL((permute_kernel<T>),                                   \
                       dim3(galv_grid((packs + 255) / 256)), dim3(256), 0, s, \
                       x, y, rows, m, h);                                     \
  }                                                                           \
  void moe_permute_bwd_launch_##T(const T* dy, float* acc, const long* rows,  \
                                  long m, int h, hipStream_t s) {             \
    long packs = m * (h / 8);                                                 \
    hipLaunchKernelGGL((permute_bwd_kernel<T>),                               \
                       dim3(galv_grid((packs + 255) / 256)), dim3(256), 0, s, \
                       dy, acc, rows, m, h);                                  \
  }                                                                           \
  void moe_unpermute_launch_##T(const T* back, const float* probs,            \
                                const long* inv, T* out, long n, int k,       \
                                int h, hipStream_t s) {                       \
    long packs = n * (h / 8);                                                 \
    hipLaunchKernelGGL((unpermute_kernel<T>),                                 \
                       dim3(galv_grid((packs + 255) / 256)), dim3(256), 0, s, \
                       back, probs, inv, out, n, k, h);                       \
  }                                                                           \
  void moe_unpermute_bwd_launch_##T(const T* dout, const T* back,             \
                                    const float* probs, const long* t_of,     \
                                    T* dback, float* dprobs, long m, int h,   \
                                    hipStream_t s) {                          \
    long packs = m * (h / 8);                                                 \
    hipLaunchKernelGGL((unpermute_bwd_kernel<T>),                             \
                       dim3(galv_grid((packs + 255) / 256)), dim3(256), 0, s, \
                       dout, back, probs, t_of, dback, dprobs, m, h);         \
  }

typedef __bf16 bf16_t;
typedef float f32_t;
DEF_LAUNCH(bf16_t)
DEF_LAUNCH(f32_t)
