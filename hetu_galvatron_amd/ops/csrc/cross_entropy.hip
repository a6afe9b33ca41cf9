// Vocab-parallel cross-entropy shard-local passes for gfx950.
//
// Replaces the reference's Triton tiled CE
// (galvatron/core/runtime/tensor_parallel/triton_cross_entropy.py:21-256):
// three kernels — row max, sum-exp + target-logit pick, in-place softmax
// gradient — with the TP all-reduces of [n]-shaped fp32 stats done by the
// host wrapper (runtime/tensor_parallel/cross_entropy.py).
//
// logits: [n, V] bf16|fp32; V is the LOCAL vocab shard width (any value,
// vectorized main loop + scalar tail). Backward overwrites the logits
// buffer in place (the [n,V] logits tensor dominates lm-head activation
// memory at 128k vocab).
#include "common.h"

namespace {

template <typename T>
__global__ void ce_max_kernel(const T* __restrict__ logits,
                              float* __restrict__ out, long n, long V) {
  __shared__ float red[4];
  const int tid = threadIdx.x;
  const long V8 = V & ~7L;
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const T* lr = logits + row * V;
    float mx = -3.4e38f;
    for (long c = tid * 8; c < V8; c += 256 * 8) {
      float v[8];
      VecIO<T>::load(v, lr + c);
#pragma unroll
      for (int i = 0; i < 8; ++i) mx = fmaxf(mx, v[i]);
    }
    for (long c = V8 + tid; c < V; c += 256) mx = fmaxf(mx, (float)lr[c]);
    mx = block_max_256(mx, red);
    if (tid == 0) out[row] = mx;
  }
}

template <typename T>
__global__ void ce_sum_target_kernel(const T* __restrict__ logits,
                                     const long* __restrict__ target,
                                     const float* __restrict__ gmax,
                                     float* __restrict__ sumexp,
                                     float* __restrict__ tlogit,
                                     long n, long V, long vocab_start) {
  __shared__ float red[4];
  const int tid = threadIdx.x;
  const long V8 = V & ~7L;
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const T* lr = logits + row * V;
    const float mx = gmax[row];
    const long t_local = target[row] - vocab_start;
    float se = 0.f;
    float tl = 0.f;
    for (long c = tid * 8; c < V8; c += 256 * 8) {
      float v[8];
      VecIO<T>::load(v, lr + c);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        se += __expf(v[i] - mx);
        if (c + i == t_local) tl = v[i];
      }
    }
    for (long c = V8 + tid; c < V; c += 256) {
      const float v = (float)lr[c];
      se += __expf(v - mx);
      if (c == t_local) tl = v;
    }
    se = block_sum_256(se, red);
    // exactly one thread saw the target column (or none, off-shard);
    // a sum-reduce broadcasts its value.
    tl = block_sum_256(tl, red);
    if (tid == 0) {
      sumexp[row] = se;
      tlogit[row] = (t_local >= 0 && t_local < V) ? tl : 0.f;
    }
  }
}

template <typename T>
__global__ void ce_bwd_kernel(T* __restrict__ logits,
                              const long* __restrict__ target,
                              const float* __restrict__ gmax,
                              const float* __restrict__ sumexp,
                              const float* __restrict__ gout,
                              long n, long V, long vocab_start) {
  const int tid = threadIdx.x;
  const long V8 = V & ~7L;
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    T* lr = logits + row * V;
    const float mx = gmax[row];
    const float inv_se = 1.f / sumexp[row];
    const float g = gout[row];
    const long t_local = target[row] - vocab_start;
    for (long c = tid * 8; c < V8; c += 256 * 8) {
      float v[8];
      VecIO<T>::load(v, lr + c);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float p = __expf(v[i] - mx) * inv_se;
        if (c + i == t_local) p -= 1.f;
        v[i] = p * g;
      }
      VecIO<T>::store(lr + c, v);
    }
    for (long c = V8 + tid; c < V; c += 256) {
      float p = __expf((float)lr[c] - mx) * inv_se;
      if (c == t_local) p -= 1.f;
      lr[c] = (T)(p * g);
    }
  }
}

}  // namespace

template <typename T>
void ce_max_launch_t(const T* logits, float* out, long n, long V,
                     hipStream_t st) {
  hipLaunchKernelGGL((ce_max_kernel<T>), dim3(galv_grid(n)), dim3(256), 0,
                     st, logits, out, n, V);
}

template <typename T>
void ce_sum_target_launch_t(const T* logits, const long* target,
                            const float* gmax, float* sumexp, float* tlogit,
                            long n, long V, long vocab_start, hipStream_t st) {
  hipLaunchKernelGGL((ce_sum_target_kernel<T>), dim3(galv_grid(n)), dim3(256),
                     0, st, logits, target, gmax, sumexp, tlogit, n, V,
                     vocab_start);
}

template <typename T>
void ce_bwd_launch_t(T* logits, const long* target, const float* gmax,
                     const float* sumexp, const float* gout, long n, long V,
                     long vocab_start, hipStream_t st) {
  hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(galv_grid(n)), dim3(256), 0,
                     st, logits, target, gmax, sumexp, gout, n, V,
                     vocab_start);
}

template void ce_max_launch_t<__bf16>(const __bf16*, float*, long, long, hipStream_t);
template void ce_max_launch_t<float>(const float*, float*, long, long, hipStream_t);
template void ce_sum_target_launch_t<__bf16>(const __bf16*, const long*, const float*, float*, float*, long, long, long, hipStream_t);
template void ce_sum_target_launch_t<float>(const float*, const long*, const float*, float*, float*, long, long, long, hipStream_t);
template void ce_bwd_launch_t<__bf16>(__bf16*, const long*, const float*, const float*, const float*, long, long, long, hipStream_t);
template void ce_bwd_launch_t<float>(float*, const long*, const float*, const float*, const float*, long, long, long, hipStream_t);
