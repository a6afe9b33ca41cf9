#include "hip/hip_runtime.h"
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

// ---------------------------------------------------------------------------
// Single-token decode attention (serving): o[b,hq,d] from a KV cache.
// Reference role: nvidia_chunked_flash_attn flash-decode
// (attention.py:398-514, optional dep).  Memory-bound KV read (guide
// App. B "Attention decode"): one 256-thread block per (b, hq);
// phase 1: lanes stripe cache positions, 16 B vectorized K dot + online
// softmax, weights staged in LDS; phase 2: threads stripe d, stream V.
// ---------------------------------------------------------------------------
namespace {

template <int D>
__global__ void decode_attn_kernel(const __bf16* __restrict__ q,   // [b,hq,D]
                                   const __bf16* __restrict__ kc,  // [b,S,hkv,D]
                                   const __bf16* __restrict__ vc,
                                   __bf16* __restrict__ o,         // [b,hq,D]
                                   int b, int hq, int hkv, int max_s,
                                   int cur_len, float scale) {
  extern __shared__ __align__(16) float wsm[];  // [cur_len] weights
  __shared__ float red[2][8];
  const int bh = blockIdx.x;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);
  const int tid = threadIdx.x;

  // q into registers per thread chunk? phase 1 needs full q per lane: LDS
  __shared__ __align__(16) float q_sh[D];
  if (tid < D / 8) {
    float v[8];
    VecIO<__bf16>::load(v, q + ((long)bi * hq + h) * D + tid * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) q_sh[tid * 8 + i] = v[i];
  }
  __syncthreads();

  const long kbase = ((long)bi * max_s * hkv + hk) * D;
  const int kstride = hkv * D;

  float m = -1e30f, l = 0.f;
  for (int p = tid; p < cur_len; p += blockDim.x) {
    const __bf16* kp = kc + kbase + (long)p * kstride;
    float dot = 0.f;
#pragma unroll
    for (int c = 0; c < D; c += 8) {
      float kv[8];
      VecIO<__bf16>::load(kv, kp + c);
#pragma unroll
      for (int i = 0; i < 8; ++i) dot += kv[i] * q_sh[c + i];
    }
    const float s = dot * scale;
    wsm[p] = s;
    if (s > m) {
      l *= __expf(m - s);
      m = s;
    }
    l += __expf(s - m);
  }
  // merge (m, l) across the block
  const int lane = tid & 63, wid = tid >> 6;
#pragma unroll
  for (int sft = 32; sft >= 1; sft >>= 1) {
    float mo = __shfl_xor(m, sft, 64);
    float lo = __shfl_xor(l, sft, 64);
    float mn = fmaxf(m, mo);
    l = l * __expf(m - mn) + lo * __expf(mo - mn);
    m = mn;
  }
  if (lane == 0) { red[0][wid] = m; red[1][wid] = l; }
  __syncthreads();
  float gm = -1e30f, gl = 0.f;
  for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
    float mn = fmaxf(gm, red[0][w]);
    gl = gl * __expf(gm - mn) + red[1][w] * __expf(red[0][w] - mn);
    gm = mn;
  }
  __syncthreads();
  // normalize weights in LDS
  for (int p = tid; p < cur_len; p += blockDim.x)
    wsm[p] = __expf(wsm[p] - gm) / gl;
  __syncthreads();

  // phase 2: stream V with 16 B vector loads — thread owns one 8-elem
  // d-octet, positions striped over 2048/D slices, LDS tree at the end.
  constexpr int OCTS = D / 8;        // octets per row
  constexpr int PSL = 256 / OCTS;    // position slices
  __shared__ float acc_sh[256][8];
  const int oct = tid % OCTS, sl = tid / OCTS;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int p = sl; p < cur_len; p += PSL) {
    const float w = wsm[p];
    float vv[8];
    VecIO<__bf16>::load(vv, vc + kbase + (long)p * kstride + oct * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += w * vv[j];
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) acc_sh[sl * OCTS + oct][j] = acc[j];
  __syncthreads();
  if (tid < D) {
    const int od = tid / 8, j = tid % 8;
    float s2 = 0.f;
#pragma unroll 4
    for (int s3 = 0; s3 < PSL; ++s3) s2 += acc_sh[s3 * OCTS + od][j];
    o[((long)bi * hq + h) * D + tid] = (__bf16)s2;
  }
}

// Split-KV variant: small-batch decode launches only b*hq blocks, which
// starves 256 CUs (measured 50 GB/s at b=1).  Stripe the cache over
// n_chunks extra grid-y blocks; each writes an UNNORMALIZED partial
// (o_c = sum exp(s-m_c) v, plus m_c, l_c) and a merge kernel combines
// them with the usual log-sum-exp reweighting (flash-decode scheme).
template <int D>
__global__ void decode_attn_split_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ kc,
    const __bf16* __restrict__ vc,
    float* __restrict__ part,  // [b*hq, n_chunks, D+2]
    int b, int hq, int hkv, int max_s, int cur_len, int chunk, float scale,
    const int* __restrict__ cur_len_dev) {  // non-null: hipGraph mode, the
                                            // length lives in device memory
  extern __shared__ __align__(16) float wsm[];  // [chunk] weights
  __shared__ float red[2][8];
  if (cur_len_dev != nullptr) {
    cur_len = *cur_len_dev;
    chunk = (cur_len + gridDim.y - 1) / gridDim.y;
  }
  const int bh = blockIdx.x;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);
  const int tid = threadIdx.x;
  const int p0 = blockIdx.y * chunk;
  const int p1 = min(p0 + chunk, cur_len);
  float* out = part + ((long)bh * gridDim.y + blockIdx.y) * (D + 2);
  if (p0 >= p1) {  // empty tail chunk
    for (int d0 = tid; d0 < D; d0 += blockDim.x) out[d0] = 0.f;
    if (tid == 0) { out[D] = -1e30f; out[D + 1] = 0.f; }
    return;
  }

  __shared__ __align__(16) float q_sh[D];
  if (tid < D / 8) {
    float v[8];
    VecIO<__bf16>::load(v, q + ((long)bi * hq + h) * D + tid * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) q_sh[tid * 8 + i] = v[i];
  }
  __syncthreads();

  const long kbase = ((long)bi * max_s * hkv + hk) * D;
  const int kstride = hkv * D;

  float m = -1e30f, l = 0.f;
  for (int p = p0 + tid; p < p1; p += blockDim.x) {
    const __bf16* kp = kc + kbase + (long)p * kstride;
    float dot = 0.f;
#pragma unroll
    for (int c = 0; c < D; c += 8) {
      float kv[8];
      VecIO<__bf16>::load(kv, kp + c);
#pragma unroll
      for (int i = 0; i < 8; ++i) dot += kv[i] * q_sh[c + i];
    }
    const float s = dot * scale;
    wsm[p - p0] = s;
    if (s > m) {
      l *= __expf(m - s);
      m = s;
    }
    l += __expf(s - m);
  }
  const int lane = tid & 63, wid = tid >> 6;
#pragma unroll
  for (int sft = 32; sft >= 1; sft >>= 1) {
    float mo = __shfl_xor(m, sft, 64);
    float lo = __shfl_xor(l, sft, 64);
    float mn = fmaxf(m, mo);
    l = l * __expf(m - mn) + lo * __expf(mo - mn);
    m = mn;
  }
  if (lane == 0) { red[0][wid] = m; red[1][wid] = l; }
  __syncthreads();
  float gm = -1e30f, gl = 0.f;
  for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
    float mn = fmaxf(gm, red[0][w]);
    gl = gl * __expf(gm - mn) + red[1][w] * __expf(red[0][w] - mn);
    gm = mn;
  }
  __syncthreads();
  for (int p = tid; p < p1 - p0; p += blockDim.x)
    wsm[p] = __expf(wsm[p] - gm);  // unnormalized: merge divides by gl
  __syncthreads();

  constexpr int OCTS = D / 8;
  constexpr int PSL = 256 / OCTS;
  __shared__ float acc_sh[256][8];
  const int oct = tid % OCTS, sl = tid / OCTS;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int p = p0 + sl; p < p1; p += PSL) {
    const float w = wsm[p - p0];
    float vv[8];
    VecIO<__bf16>::load(vv, vc + kbase + (long)p * kstride + oct * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += w * vv[j];
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) acc_sh[sl * OCTS + oct][j] = acc[j];
  __syncthreads();
  if (tid < D) {
    const int od = tid / 8, j = tid % 8;
    float s2 = 0.f;
#pragma unroll 4
    for (int s3 = 0; s3 < PSL; ++s3) s2 += acc_sh[s3 * OCTS + od][j];
    out[tid] = s2;
  }
  if (tid == 0) { out[D] = gm; out[D + 1] = gl; }
}

template <int D>
__global__ void decode_attn_merge_kernel(const float* __restrict__ part,
                                         __bf16* __restrict__ o, int n_chunks) {
  const int bh = blockIdx.x;
  const int tid = threadIdx.x;  // blockDim.x == D
  const float* base = part + (long)bh * n_chunks * (D + 2);
  float gm = -1e30f;
  for (int c = 0; c < n_chunks; ++c) gm = fmaxf(gm, base[c * (D + 2) + D]);
  float gl = 0.f, acc = 0.f;
  for (int c = 0; c < n_chunks; ++c) {
    const float w = __expf(base[c * (D + 2) + D] - gm);
    gl += w * base[c * (D + 2) + D + 1];
    acc += w * base[c * (D + 2) + tid];
  }
  o[(long)bh * D + tid] = (__bf16)(acc / gl);
}

}  // namespace

void decode_attn_launch(const __bf16* q, const __bf16* kc, const __bf16* vc,
                        __bf16* o, float* part_ws, int n_chunks, int b,
                        int hq, int hkv, int max_s, int cur_len, int d,
                        float scale, const int* cur_len_dev, hipStream_t st) {
  if (n_chunks <= 1 && cur_len_dev == nullptr) {
    const int shmem = cur_len * sizeof(float);
    if (d == 64)
      hipLaunchKernelGGL((decode_attn_kernel<64>), dim3(b * hq), dim3(256),
                         shmem, st, q, kc, vc, o, b, hq, hkv, max_s, cur_len,
                         scale);
    else
      hipLaunchKernelGGL((decode_attn_kernel<128>), dim3(b * hq), dim3(256),
                         shmem, st, q, kc, vc, o, b, hq, hkv, max_s, cur_len,
                         scale);
    return;
  }
  // graph mode sizes the LDS weight buffer for the worst case (cur_len
  // unknown at launch); chunk passed is then the max over the capture
  const int chunk = (cur_len + n_chunks - 1) / n_chunks;
  const int shmem = chunk * sizeof(float);
  dim3 grid(b * hq, n_chunks);
  if (d == 64) {
    hipLaunchKernelGGL((decode_attn_split_kernel<64>), grid, dim3(256),
                       shmem, st, q, kc, vc, part_ws, b, hq, hkv, max_s,
                       cur_len, chunk, scale, cur_len_dev);
    hipLaunchKernelGGL((decode_attn_merge_kernel<64>), dim3(b * hq),
                       dim3(64), 0, st, part_ws, o, n_chunks);
  } else {
    hipLaunchKernelGGL((decode_attn_split_kernel<128>), grid, dim3(256),
                       shmem, st, q, kc, vc, part_ws, b, hq, hkv, max_s,
                       cur_len, chunk, scale, cur_len_dev);
    hipLaunchKernelGGL((decode_attn_merge_kernel<128>), dim3(b * hq),
                       dim3(128), 0, st, part_ws, o, n_chunks);
  }
}
