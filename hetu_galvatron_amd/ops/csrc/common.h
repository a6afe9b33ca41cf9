// Shared device helpers for the CDNA4 (gfx950) kernel suite.
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//  - wave = 64 lanes (hard-coded; CDNA, not a CUDA warp)
//  - bf16 memory traffic always vectorized as 8-element (16 B/lane) packets
//  - MFMA fragment lane->element maps centralized here (single source of
//    truth; validated on-device by the mfma probe in flash_attn.hip)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;

// ---------------------------------------------------------------------------
// dtype conversion (bf16 <-> f32). __bf16 converts natively in amdclang.
// ---------------------------------------------------------------------------
DEV_INLINE float bf2f(__bf16 v) { return (float)v; }
DEV_INLINE __bf16 f2bf(float v) { return (__bf16)v; }

struct BF16Tag {};
struct F32Tag {};

template <typename T> struct VecIO;

// 8-element packet of bf16 == 16 bytes (one dwordx4 load/store).
template <> struct VecIO<__bf16> {
  using pack_t = bf16x8;
  static constexpr int width = 8;
  DEV_INLINE static void load(float* dst, const __bf16* p) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(p);
#pragma unroll
    for (int i = 0; i < 8; ++i) dst[i] = (float)v[i];
  }
  DEV_INLINE static void store(__bf16* p, const float* src) {
    bf16x8 v;
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = (__bf16)src[i];
    *reinterpret_cast<bf16x8*>(p) = v;
  }
};

// 8-element packet of f32 == two dwordx4.
template <> struct VecIO<float> {
  using pack_t = f32x4;
  static constexpr int width = 8;
  DEV_INLINE static void load(float* dst, const float* p) {
    f32x4 a = *reinterpret_cast<const f32x4*>(p);
    f32x4 b = *reinterpret_cast<const f32x4*>(p + 4);
#pragma unroll
    for (int i = 0; i < 4; ++i) { dst[i] = a[i]; dst[i + 4] = b[i]; }
  }
  DEV_INLINE static void store(float* p, const float* src) {
    f32x4 a, b;
#pragma unroll
    for (int i = 0; i < 4; ++i) { a[i] = src[i]; b[i] = src[i + 4]; }
    *reinterpret_cast<f32x4*>(p) = a;
    *reinterpret_cast<f32x4*>(p + 4) = b;
  }
};

// ---------------------------------------------------------------------------
// Reductions. Wave (64-lane) shuffle tree + LDS across the block's waves.
// ---------------------------------------------------------------------------
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int m = 32; m >= 1; m >>= 1) v += __shfl_xor(v, m, WAVE);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int m = 32; m >= 1; m >>= 1) v = fmaxf(v, __shfl_xor(v, m, WAVE));
  return v;
}

// Sum across a 32-lane half-group (lanes l and l^32 stay independent).
DEV_INLINE float half_wave_sum(float v) {
#pragma unroll
  for (int m = 16; m >= 1; m >>= 1) v += __shfl_xor(v, m, WAVE);
  return v;
}

DEV_INLINE float half_wave_max(float v) {
#pragma unroll
  for (int m = 16; m >= 1; m >>= 1) v = fmaxf(v, __shfl_xor(v, m, WAVE));
  return v;
}

// Block-level reduce for 256-thread (4-wave) blocks. `red` is a 4-float LDS
// scratch. Returns the reduced value broadcast to every thread.
DEV_INLINE float block_sum_256(float v, float* red) {
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = v;
  __syncthreads();
  float r = red[0] + red[1] + red[2] + red[3];
  __syncthreads();
  return r;
}

DEV_INLINE float block_max_256(float v, float* red) {
  const int wid = threadIdx.x / WAVE;
  v = wave_max(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = v;
  __syncthreads();
  float r = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  __syncthreads();
  return r;
}

// ---------------------------------------------------------------------------
// MFMA fragment lane maps: v_mfma_f32_32x32x16_bf16 (M=N=32, K=16).
//   A[32x16]: lane l holds A[l&31][ 8*(l>>5) + j ], j = 0..7   (contiguous-K)
//   B[16x32]: lane l holds B[ 8*(l>>5) + j ][l&31]
//   D[32x32]: reg r (0..15): D[(r&3) + 8*(r>>2) + 4*(l>>5)][l&31]
// The A/B map is validated on-device by mfma_probe (flash_attn.hip); the
// alternative two-block-of-4 map is probe-selectable for verification.
// ---------------------------------------------------------------------------
DEV_INLINE int mfma32_ab_k(int lane, int j) { return 8 * (lane >> 5) + j; }
DEV_INLINE int mfma32_ab_k_alt(int lane, int j) {
  // two K=8 blocks, CDNA3-style 4-element groups inside each
  return (j < 4) ? (4 * (lane >> 5) + j) : (8 + 4 * (lane >> 5) + (j - 4));
}
DEV_INLINE int mfma32_d_row(int lane, int r) {
  return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
}
DEV_INLINE int mfma32_d_col(int lane) { return lane & 31; }

DEV_INLINE f32x16 mfma32_bf16(bf16x8 a, bf16x8 b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// Grid sizing: memory-bound kernels cap at 8 blocks x 256 CUs and
// grid-stride the rest (guideline 11).
// ---------------------------------------------------------------------------
inline int galv_grid(long total_blocks, int cap = 2048) {
  return (int)(total_blocks < cap ? total_blocks : cap);
}

#define HIP_CHECK(cmd)                                                     \
  do {                                                                     \
    hipError_t e_ = (cmd);                                                 \
    if (e_ != hipSuccess) {                                                \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e_), __FILE__,   \
             __LINE__);                                                    \
    }                                                                      \
  } while (0)
