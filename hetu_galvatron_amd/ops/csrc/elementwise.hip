// Fused SwiGLU / GeGLU activation and rotary position embedding for gfx950.
//
// Replaces the reference's torch.jit fused bias-activations
// (galvatron/core/runtime/transformer/fused_kernels.py:143-226) and the
// flash-attn rotary_emb CUDA extension (fused_kernels.py:227-257).
//
// Memory-bound elementwise: vectorized 16 B/lane bf16 packets, grid-stride.
// RoPE consumes HOST-precomputed cos/sin tables (guide App. B: no on-device
// trig in the hot path); NEOX half-rotation layout, x: [s, b, h, d].
#include "common.h"

namespace {

template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  long rows, int F) {
  const long packs_per_row = F / 8;
  const long total = rows * packs_per_row;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / packs_per_row;
    const long c = (idx - row * packs_per_row) * 8;
    const T* g = x + row * (long)(2 * F) + c;
    const T* u = g + F;
    float gv[8], uv[8], out[8];
    VecIO<T>::load(gv, g);
    VecIO<T>::load(uv, u);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float sig = 1.f / (1.f + __expf(-gv[i]));
      out[i] = gv[i] * sig * uv[i];
    }
    VecIO<T>::store(y + row * (long)F + c, out);
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dy,
                                  const T* __restrict__ x,
                                  T* __restrict__ dx, long rows, int F) {
  const long packs_per_row = F / 8;
  const long total = rows * packs_per_row;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / packs_per_row;
    const long c = (idx - row * packs_per_row) * 8;
    const T* g = x + row * (long)(2 * F) + c;
    const T* u = g + F;
    float gv[8], uv[8], dyv[8], dg[8], du[8];
    VecIO<T>::load(gv, g);
    VecIO<T>::load(uv, u);
    VecIO<T>::load(dyv, dy + row * (long)F + c);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float sig = 1.f / (1.f + __expf(-gv[i]));
      const float silu = gv[i] * sig;
      dg[i] = dyv[i] * uv[i] * sig * (1.f + gv[i] * (1.f - sig));
      du[i] = dyv[i] * silu;
    }
    T* dgp = dx + row * (long)(2 * F) + c;
    VecIO<T>::store(dgp, dg);
    VecIO<T>::store(dgp + F, du);
  }
}

// x: [s, b, h, d] -> rows = s*b*h of length d; cos/sin: [s, d/2] fp32.
// NEOX: y1 = x1*c - x2*s ; y2 = x2*c + x1*s  (conj flips the sign of s).
template <typename T, bool CONJ>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            long rows, int bh, int d) {
  const int d2 = d / 2;
  const long packs_per_row = d2 / 8;
  const long total = rows * packs_per_row;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / packs_per_row;
    const long j = (idx - row * packs_per_row) * 8;
    const long s_idx = row / bh;
    const T* x1 = x + row * (long)d + j;
    const T* x2 = x1 + d2;
    float a[8], b[8], c[8], s[8], o1[8], o2[8];
    VecIO<T>::load(a, x1);
    VecIO<T>::load(b, x2);
    VecIO<float>::load(c, cos_t + s_idx * d2 + j);
    VecIO<float>::load(s, sin_t + s_idx * d2 + j);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float sv = CONJ ? -s[i] : s[i];
      o1[i] = a[i] * c[i] - b[i] * sv;
      o2[i] = b[i] * c[i] + a[i] * sv;
    }
    T* y1 = y + row * (long)d + j;
    VecIO<T>::store(y1, o1);
    VecIO<T>::store(y1 + d2, o2);
  }
}

}  // namespace

template <typename T>
void swiglu_fwd_launch_t(const T* x, T* y, long rows, int F, hipStream_t st) {
  long total = rows * (F / 8);
  int grid = galv_grid((total + 255) / 256);
  hipLaunchKernelGGL((swiglu_fwd_kernel<T>), dim3(grid), dim3(256), 0, st,
                     x, y, rows, F);
}

template <typename T>
void swiglu_bwd_launch_t(const T* dy, const T* x, T* dx, long rows, int F,
                         hipStream_t st) {
  long total = rows * (F / 8);
  int grid = galv_grid((total + 255) / 256);
  hipLaunchKernelGGL((swiglu_bwd_kernel<T>), dim3(grid), dim3(256), 0, st,
                     dy, x, dx, rows, F);
}

template <typename T>
void rope_launch_t(const T* x, T* y, const float* cos_t, const float* sin_t,
                   long rows, int bh, int d, bool conj, hipStream_t st) {
  long total = rows * (d / 2 / 8);
  int grid = galv_grid((total + 255) / 256);
  if (conj)
    hipLaunchKernelGGL((rope_kernel<T, true>), dim3(grid), dim3(256), 0, st,
                       x, y, cos_t, sin_t, rows, bh, d);
  else
    hipLaunchKernelGGL((rope_kernel<T, false>), dim3(grid), dim3(256), 0, st,
                       x, y, cos_t, sin_t, rows, bh, d);
}

template void swiglu_fwd_launch_t<__bf16>(const __bf16*, __bf16*, long, int, hipStream_t);
template void swiglu_fwd_launch_t<float>(const float*, float*, long, int, hipStream_t);
template void swiglu_bwd_launch_t<__bf16>(const __bf16*, const __bf16*, __bf16*, long, int, hipStream_t);
template void swiglu_bwd_launch_t<float>(const float*, const float*, float*, long, int, hipStream_t);
template void rope_launch_t<__bf16>(const __bf16*, __bf16*, const float*, const float*, long, int, int, bool, hipStream_t);
template void rope_launch_t<float>(const float*, float*, const float*, const float*, long, int, int, bool, hipStream_t);
