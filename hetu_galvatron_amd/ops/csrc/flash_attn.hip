// CDNA4 MFMA flash attention (forward with LSE, backward) for gfx950.
//
// Replaces the reference's FlashAttention-2 CUDA dependency
// (galvatron/core/runtime/transformer/attention_impl.py:18-108 and the raw
// _flash_attn_forward/_backward used by ring attention at :561-729).
// LSE is a first-class output (ring-CP merge needs it).
//
// Layout: q [b, sq, hq, D], k/v [b, skv, hkv, D] bf16, GQA (hq % hkv == 0),
// causal = bottom-right aligned. lse [b, hq, sq] fp32 (natural log).
//
// v2 forward structure (guide §B 8-warp ladder, plain HIP):
//   8 waves x 32 q rows (QB=256/workgroup), KV tiles of 64;
//   swapped QK^T (S^T = mfma(K, Q^T)) so each lane owns one q column ->
//   softmax fully in-register (per-lane m/l scalars, exp2 units);
//   P redistributed to PV fragments by cvt_pk_bf16 + permlane32_swap
//   (no LDS round-trip); PV computed as O^T = mfma(V^T, P^T) so the
//   running rescale is a plain per-lane multiply;
//   V staged TRANSPOSED straight from global (coalesced 2-byte column
//   loads -> b128 LDS writes; the v1 scalar ds_write_b16 transpose was
//   14-37% of wave cycles in SQ_LDS_BANK_CONFLICT);
//   async-stage split (T14): next tile's global loads issue before the
//   current tile's compute, LDS writes land after the barrier;
//   per-wave causal tile skip.
// Backward keeps the v1 two-kernel (dq + expanded dkv) structure with the
// same column-load transpose staging.
#include "common.h"

namespace {

constexpr int KVB = 32;   // kv tile of the BACKWARD kernels
constexpr int QB = 128;   // q/key rows per workgroup (backward kernels)

DEV_INLINE float neg_big() { return -1e30f; }

DEV_INLINE unsigned pack_bf16(float lo, float hi) {
  union { __hip_bfloat162 h2; unsigned u; } cv;
  cv.h2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
  return cv.u;
}

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// 16 bias values for one 32-key D-layout column: row(r) = 8*(r>>2) +
// 4*hi + (r&3) — each r-quad is 4 contiguous bf16, so 4 8-byte loads
// cover the column.  base must be 4-element aligned (callers pass
// kv0 + 4*hi with kv0 % 32 == 0); the clamp keeps the tail in bounds
// (masked lanes discard the values).
DEV_INLINE void load_bias16(const __bf16* brow, int base, int limit,
                            float* out /*16*/) {
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    int idx = base + 8 * g;
    idx = min(idx, max((limit - 4) & ~3, 0));
    const bf16x4 b4 = *reinterpret_cast<const bf16x4*>(brow + idx);
#pragma unroll
    for (int j = 0; j < 4; ++j) out[4 * g + j] = (float)b4[j];
  }
}

// ---------------------------------------------------------------------------
// forward (v2)
// ---------------------------------------------------------------------------
template <int D, bool BIASED = false, bool WINDOWED = false>
__device__ __attribute__((noinline))
void flash_fwd_block(int qblk, const __bf16* __restrict__ q,
                     const __bf16* __restrict__ k,
                     const __bf16* __restrict__ v,
                     __bf16* __restrict__ o, float* __restrict__ lse,
                     __bf16* smem_base, long q_base, long kv_base,
                     long lse_base, int q_stride, int kv_stride, int off,
                     int sq, int skv, float sl2e, bool causal,
                     const __bf16* __restrict__ bias = nullptr,
                     long bias_base = 0, int window = 0) {
  constexpr int KB = 64;            // kv tile
  constexpr int QBF = 256;          // q rows per workgroup (8 waves x 32)
  constexpr int KROW = D + 8;       // padded K row (bf16 elems)
  constexpr int VTROW = KB + 8;     // padded transposed-V row
  constexpr int NK = D / 16;        // S^T K-slices
  constexpr int ND = D / 32;        // 32-wide d tiles of O^T
  constexpr int KPT = D / 64;       // staging packs per thread (512 thr)

  constexpr int BUFSZ = KB * KROW + D * VTROW;
  __bf16* smem = smem_base;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int col = lane & 31;
  const int hi = lane >> 5;
  const int wid = tid >> 6;
  const int q0w = qblk * QBF + wid * 32;

  // Q fragments (B-operand of S^T): lane holds Q[q0w+col][ks*16 + 8*hi + j]
  bf16x8 qf[NK];
  {
    const int qg = min(q0w + col, sq - 1);
    const __bf16* qp = q + q_base + (long)qg * q_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks)
      qf[ks] = *reinterpret_cast<const bf16x8*>(qp + ks * 16);
  }

  f32x16 ot[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) ot[dt] = (f32x16)(0.f);
  float m_run = neg_big(), l_run = 0.f;

  int kv_end = causal ? min(skv, qblk * QBF + QBF + off) : skv;
  const int kv_last_w = causal ? min(skv, q0w + 32 + off) : skv;  // wave's own
  // sliding window (mistral): keys visible to q are [q+off-window+1, q+off];
  // whole-workgroup start tile + per-wave lower skip + per-element mask
  int kv_begin = 0;
  int kv_first_w = 0;
  if (WINDOWED) {
    kv_begin = max(0, (qblk * QBF + off - window + 1) / KB * KB);
    kv_first_w = max(0, q0w + off - window + 1);
  }

  // staged registers for the next tile
  bf16x8 kst[KPT];
  ushort8 vst[KPT];

  // Guards are hoisted out of the load loops: per-element branches around
  // staged loads make hipcc wait vmcnt(0) after EACH load (32 serialized
  // memory round trips per tile, ~13 us/tile measured — guide §5 trap (c)).
  // Fast path is branch-free; the tail tile loads clamped addresses and
  // zeroes by value-select.
  auto stage_load = [&](int kv0) {
    const bool full = (kv0 + KB <= skv);
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      const int kg = kv0 + row;
      if (full) {
        kst[p] = *reinterpret_cast<const bf16x8*>(
            k + kv_base + (long)kg * kv_stride + c8);
      } else {
        bf16x8 t = *reinterpret_cast<const bf16x8*>(
            k + kv_base + (long)min(kg, skv - 1) * kv_stride + c8);
        kst[p] = kg < skv ? t : (bf16x8)(__bf16(0.f));
      }
    }
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      const int c = idx & (D - 1);
      const int kc = (idx / D) * 8;
      const unsigned short* vb = reinterpret_cast<const unsigned short*>(
          v + kv_base) + c;
      ushort8 vv;
      if (full) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vv[j] = vb[(long)(kv0 + kc + j) * kv_stride];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned short t = vb[(long)min(kv0 + kc + j, skv - 1) * kv_stride];
          vv[j] = kv0 + kc + j < skv ? t : (unsigned short)0;
        }
      }
      vst[p] = vv;
    }
  };
  auto stage_write = [&](int buf) {
    __bf16* k_lds = smem + buf * BUFSZ;
    __bf16* vt_lds = k_lds + KB * KROW;
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      *reinterpret_cast<bf16x8*>(k_lds + row * KROW + c8) = kst[p];
    }
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      const int c = idx & (D - 1);
      const int kc = (idx / D) * 8;
      *reinterpret_cast<ushort8*>(vt_lds + c * VTROW + kc) = vst[p];
    }
  };

  stage_load(kv_begin);
  stage_write(0);
  __syncthreads();
  int cur = 0;

  for (int kv0 = kv_begin; kv0 < kv_end; kv0 += KB) {
    const bool have_next = kv0 + KB < kv_end;
    const __bf16* k_lds = smem + cur * BUFSZ;
    const __bf16* vt_lds = k_lds + KB * KROW;
    if (have_next) stage_load(kv0 + KB);  // async: lands at stage_write

    if (kv0 < kv_last_w && kv0 + KB > kv_first_w) {  // per-wave skips
      // ---- S^T = K Q^T (two 32-key tiles) ----
      f32x16 st0 = (f32x16)(0.f), st1 = (f32x16)(0.f);
#pragma unroll
      for (int ks = 0; ks < NK; ++ks) {
        bf16x8 k0 = *reinterpret_cast<const bf16x8*>(
            k_lds + col * KROW + ks * 16 + 8 * hi);
        bf16x8 k1 = *reinterpret_cast<const bf16x8*>(
            k_lds + (col + 32) * KROW + ks * 16 + 8 * hi);
        st0 = mfma32_bf16(k0, qf[ks], st0);
        st1 = mfma32_bf16(k1, qf[ks], st1);
      }

      // ---- scale (+ additive bias, + mask on straddle/tail tiles) ----
      float pv[32];
      const int qg = q0w + col;
      const __bf16* brow = nullptr;
      if (BIASED)
        brow = bias + bias_base + (long)min(qg, sq - 1) * skv;
      const bool need_mask =
          (causal && kv0 + KB > q0w + off + 1) || (kv0 + KB > skv) ||
          (WINDOWED && kv0 < q0w + 32 + off - window + 1);
      constexpr float LOG2E = 1.4426950408889634f;
      float bv[32];
      if (BIASED) {
        load_bias16(brow, kv0 + 4 * hi, skv, bv);
        load_bias16(brow, kv0 + 32 + 4 * hi, skv, bv + 16);
      }
      if (need_mask) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key0 = kv0 + mfma32_d_row(lane, r);
          float x0 = st0[r] * sl2e, x1 = st1[r] * sl2e;
          if (BIASED) {
            x0 += LOG2E * bv[r];
            x1 += LOG2E * bv[16 + r];
          }
          if (key0 >= skv || (causal && key0 > qg + off) ||
              (WINDOWED && key0 <= qg + off - window)) x0 = neg_big();
          if (key0 + 32 >= skv || (causal && key0 + 32 > qg + off) ||
              (WINDOWED && key0 + 32 <= qg + off - window))
            x1 = neg_big();
          pv[r] = x0;
          pv[16 + r] = x1;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          pv[r] = st0[r] * sl2e + (BIASED ? LOG2E * bv[r] : 0.f);
          pv[16 + r] = st1[r] * sl2e + (BIASED ? LOG2E * bv[16 + r] : 0.f);
        }
      }

      // ---- in-register online softmax (per-lane q column) ----
      float mt = pv[0];
#pragma unroll
      for (int i = 1; i < 32; ++i) mt = fmaxf(mt, pv[i]);
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
      const float mn = fmaxf(m_run, mt);
      const float alpha = (m_run <= neg_big()) ? 0.f : exp2f(m_run - mn);
      const bool dead = (mn <= neg_big());
      float ps = 0.f;
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        pv[i] = dead ? 0.f : exp2f(pv[i] - mn);
        ps += pv[i];
      }
      ps += __shfl_xor(ps, 32, 64);
      l_run = l_run * alpha + ps;
      m_run = mn;
#pragma unroll
      for (int dt = 0; dt < ND; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;

      // ---- P -> bf16 PV fragments via cvt_pk + permlane32_swap ----
      // frag[ks] holds P[q=lane][16*ks + 8*hi + j], j=0..7, as 4 u32 words
      unsigned w[16];
#pragma unroll
      for (int t = 0; t < 2; ++t) {
        const float* pt = pv + 16 * t;
#pragma unroll
        for (int i = 0; i < 2; ++i) {
#pragma unroll
          for (int g = 0; g < 2; ++g) {
            // pairs (8i+2g, 8i+2g+1) with (8i+2g+4, 8i+2g+5)
            unsigned a = pack_bf16(pt[8 * i + 2 * g], pt[8 * i + 2 * g + 1]);
            unsigned bb = pack_bf16(pt[8 * i + 2 * g + 4],
                                    pt[8 * i + 2 * g + 5]);
            auto r2 = __builtin_amdgcn_permlane32_swap(a, bb, false, false);
            // words of frag (2t + i): position 2g -> r2[0], 2g+1... see map
            w[(2 * t + i) * 4 + g] = r2[0];      // keys (8hi + 2g, +1)
            w[(2 * t + i) * 4 + 2 + g] = r2[1];  // keys (8hi + 4 + 2g, +1)
          }
        }
      }

      // ---- O^T += V^T P^T ----
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        union { unsigned u[4]; bf16x8 f; } pb;
#pragma unroll
        for (int g = 0; g < 4; ++g) pb.u[g] = w[ks * 4 + g];
#pragma unroll
        for (int dt = 0; dt < ND; ++dt) {
          bf16x8 va = *reinterpret_cast<const bf16x8*>(
              vt_lds + (col + 32 * dt) * VTROW + ks * 16 + 8 * hi);
          ot[dt] = mfma32_bf16(va, pb.f, ot[dt]);
        }
      }
    }

    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O^T[d][q=lane], per-lane stats ----
  const int qg = q0w + col;
  if (qg < sq) {
    const float invl = l_run > 0.f ? 1.f / l_run : 0.f;
    __bf16* orow = o + q_base + (long)qg * q_stride;
#pragma unroll
    for (int dt = 0; dt < ND; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        orow[dt * 32 + mfma32_d_row(lane, r)] = (__bf16)(ot[dt][r] * invl);
    if (hi == 0)
      lse[lse_base + qg] =
          l_run > 0.f ? (m_run + log2f(l_run)) * 0.6931471805599453f
                      : -INFINITY;
  }
}

// thin kernel: common indexing + complementary-pair causal load balance
// (block x runs q blocks {x, nqb-1-x}: constant total KV tiles per block,
// so the makespan matches the average instead of 2x the deepest block)
template <int D, bool BIASED = false, bool WINDOWED = false>
__global__ __launch_bounds__(512, 2)
void flash_fwd_kernel(const __bf16* __restrict__ q,
                      const __bf16* __restrict__ k,
                      const __bf16* __restrict__ v,
                      __bf16* __restrict__ o, float* __restrict__ lse,
                      int b, int sq, int skv, int hq, int hkv, float scale,
                      bool causal, const __bf16* __restrict__ bias = nullptr,
                      bool paired = true, bool sbhd = false,
                      int window = 0) {
  constexpr int KB = 64, QBF = 256;
  constexpr int BUFSZ = KB * (D + 8) + D * (KB + 8);
  __shared__ __align__(16) __bf16 smem[2 * BUFSZ];
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);
  // sbhd: tensors are [s, b, h, d] (the runtime's native activation
  // layout) — only the per-(b,h) base and the s-stride change
  const long q_base = sbhd ? ((long)bi * hq + h) * D
                           : ((long)bi * sq * hq + h) * D;
  const long kv_base = sbhd ? ((long)bi * hkv + hk) * D
                            : ((long)bi * skv * hkv + hk) * D;
  const int q_str = (sbhd ? b * hq : hq) * D;
  const int kv_str = (sbhd ? b * hkv : hkv) * D;
  const long lse_base = ((long)bi * hq + h) * sq;
  const long bias_base = (long)h * sq * skv;  // bias [hq, sq, skv]
  const float sl2e = scale * 1.4426950408889634f;
  const int off = skv - sq;
  const int nqb = (sq + QBF - 1) / QBF;
  flash_fwd_block<D, BIASED, WINDOWED>(blockIdx.x, q, k, v, o, lse, smem, q_base,
                             kv_base, lse_base, q_str, kv_str, off, sq,
                             skv, sl2e, causal, bias, bias_base, window);
  const int qb2 = nqb - 1 - (int)blockIdx.x;
  if (causal && paired && qb2 > (int)blockIdx.x) {
    __syncthreads();
    flash_fwd_block<D, BIASED, WINDOWED>(qb2, q, k, v, o, lse, smem, q_base, kv_base,
                               lse_base, q_str, kv_str, off, sq, skv,
                               sl2e, causal, bias, bias_base, window);
  }
}

// ---------------------------------------------------------------------------
// Di = rowsum(dO * O)  — one wave per (b, s, h) row
// ---------------------------------------------------------------------------
template <int D>
__global__ void attn_di_kernel(const __bf16* __restrict__ dout,
                               const __bf16* __restrict__ o,
                               float* __restrict__ di, int b, int sq, int hq,
                               bool sbhd = false) {
  // row index = (bi*sq + s)*hq + h (bshd) or (s*b + bi)*hq + h (sbhd)
  const long rows = (long)b * sq * hq;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  constexpr int EPL = D / 64;  // elems per lane
  for (long row = blockIdx.x * 4 + wid; row < rows; row += (long)gridDim.x * 4) {
    const __bf16* dp = dout + row * D;
    const __bf16* op = o + row * D;
    float acc = 0.f;
#pragma unroll
    for (int i = 0; i < EPL; ++i) {
      const int c = lane * EPL + i;
      acc += (float)dp[c] * (float)op[c];
    }
    acc = wave_sum(acc);
    if (lane == 0) {
      // di layout [b, hq, sq]
      long bi, s, h;
      if (sbhd) {
        s = row / ((long)b * hq);
        const long rem = row - s * b * hq;
        bi = rem / hq;
        h = rem - bi * hq;
      } else {
        bi = row / ((long)sq * hq);
        const long rem = row - bi * sq * hq;
        s = rem / hq;
        h = rem - s * hq;
      }
      di[(bi * hq + h) * sq + s] = acc;
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ (v2): same swapped structure as the forward — 8 waves x 32 q
// rows, per-lane lse/di scalars, dS^T packed to MFMA fragments with
// cvt_pk + permlane32_swap (no dS LDS round-trip, no per-row stat arrays:
// the v1 form spilled 34 dwords/lane to scratch).
// dQ^T[d][q] += K^T dS accumulated in D-layout, epilogue like the forward.
// ---------------------------------------------------------------------------
template <int D, bool BIASED = false, bool WINDOWED = false>
__device__ __attribute__((noinline))
void flash_bwd_dq_block(int qblk, const __bf16* __restrict__ dout,
                        const __bf16* __restrict__ q,
                        const __bf16* __restrict__ k,
                        const __bf16* __restrict__ v,
                        const float* __restrict__ lse,
                        const float* __restrict__ di,
                        __bf16* __restrict__ dq, __bf16* smem_base,
                        long q_base, long kv_base, long lse_base,
                        int q_stride, int kv_stride, int off, int sq,
                        int skv, float scale, bool causal,
                        const __bf16* __restrict__ bias = nullptr,
                        long bias_base = 0, int window = 0) {
  constexpr int KB = 32;         // kv tile
  constexpr int QBF = 256;       // q rows per workgroup (8 waves x 32)
  constexpr int KROW = D + 8;
  constexpr int KTROW = KB + 8;
  constexpr int NK = D / 16;
  constexpr int ND = D / 32;
  constexpr int KPT = (KB * D / 8 + 511) / 512;  // row packs per thread

  constexpr int BUFSZ = 2 * KB * KROW + D * KTROW;
  __bf16* smem = smem_base;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int col = lane & 31;
  const int hi = lane >> 5;
  const int wid = tid >> 6;
  const int q0w = qblk * QBF + wid * 32;

  bf16x8 qf[NK], dof[NK];
  float lse_c, di_c;
  {
    const int qg = min(q0w + col, sq - 1);
    const __bf16* qp = q + q_base + (long)qg * q_stride + 8 * hi;
    const __bf16* dop = dout + q_base + (long)qg * q_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      qf[ks] = *reinterpret_cast<const bf16x8*>(qp + ks * 16);
      dof[ks] = *reinterpret_cast<const bf16x8*>(dop + ks * 16);
    }
    const float l0 = lse[lse_base + qg];
    const float d0 = di[lse_base + qg];
    const bool live = q0w + col < sq;
    lse_c = live ? l0 : INFINITY;  // exp(x - inf) = 0 -> dead rows silent
    di_c = live ? d0 : 0.f;
  }

  f32x16 dq_acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) dq_acc[dt] = (f32x16)(0.f);

  const int kv_end = causal ? min(skv, qblk * QBF + QBF + off) : skv;
  const int kv_last_w = causal ? min(skv, q0w + 32 + off) : skv;
  int kv_begin = 0, kv_first_w = 0;
  if (WINDOWED) {
    kv_begin = max(0, (qblk * QBF + off - window + 1) / KB * KB);
    kv_first_w = max(0, q0w + off - window + 1);
  }

  bf16x8 kst[KPT], vst_r[KPT];
  ushort8 ktst[KPT * 2];

  auto stage_load = [&](int kv0) {
    const bool full = (kv0 + KB <= skv);
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= KB * D / 8) break;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      const int kg = full ? kv0 + row : min(kv0 + row, skv - 1);
      kst[p] = *reinterpret_cast<const bf16x8*>(
          k + kv_base + (long)kg * kv_stride + c8);
      vst_r[p] = *reinterpret_cast<const bf16x8*>(
          v + kv_base + (long)kg * kv_stride + c8);
      if (!full && kv0 + row >= skv) {
        kst[p] = (bf16x8)(__bf16(0.f));
        vst_r[p] = (bf16x8)(__bf16(0.f));
      }
    }
#pragma unroll
    for (int p = 0; p < KPT * 2; ++p) {
      const int idx = tid + p * 512;
      if (idx >= D * KB / 8) break;
      const int c = idx & (D - 1);
      const int kc = (idx / D) * 8;
      const unsigned short* kb = reinterpret_cast<const unsigned short*>(
          k + kv_base) + c;
      ushort8 t;
      if (full) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          t[j] = kb[(long)(kv0 + kc + j) * kv_stride];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned short x = kb[(long)min(kv0 + kc + j, skv - 1) * kv_stride];
          t[j] = kv0 + kc + j < skv ? x : (unsigned short)0;
        }
      }
      ktst[p] = t;
    }
  };
  auto stage_write = [&](int buf) {
    __bf16* k_lds = smem + buf * BUFSZ;
    __bf16* v_lds = k_lds + KB * KROW;
    __bf16* kt_lds = v_lds + KB * KROW;
#pragma unroll
    for (int p = 0; p < KPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= KB * D / 8) break;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      *reinterpret_cast<bf16x8*>(k_lds + row * KROW + c8) = kst[p];
      *reinterpret_cast<bf16x8*>(v_lds + row * KROW + c8) = vst_r[p];
    }
#pragma unroll
    for (int p = 0; p < KPT * 2; ++p) {
      const int idx = tid + p * 512;
      if (idx >= D * KB / 8) break;
      const int c = idx & (D - 1);
      const int kc = (idx / D) * 8;
      *reinterpret_cast<ushort8*>(kt_lds + c * KTROW + kc) = ktst[p];
    }
  };

  stage_load(kv_begin);
  stage_write(0);
  __syncthreads();
  int cur = 0;

  const float sl2e = scale;  // bwd stays in natural-log units (lse is ln)

  for (int kv0 = kv_begin; kv0 < kv_end; kv0 += KB) {
    const bool have_next = kv0 + KB < kv_end;
    const __bf16* k_lds = smem + cur * BUFSZ;
    const __bf16* v_lds = k_lds + KB * KROW;
    const __bf16* kt_lds = v_lds + KB * KROW;
    if (have_next) stage_load(kv0 + KB);

    if (kv0 < kv_last_w && kv0 + KB > kv_first_w) {
      // S^T = K Q^T ; dP^T = V dO^T
      f32x16 st = (f32x16)(0.f), dp = (f32x16)(0.f);
#pragma unroll
      for (int ks = 0; ks < NK; ++ks) {
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(
            k_lds + col * KROW + ks * 16 + 8 * hi);
        st = mfma32_bf16(kb, qf[ks], st);
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            v_lds + col * KROW + ks * 16 + 8 * hi);
        dp = mfma32_bf16(vb, dof[ks], dp);
      }

      // dS^T = P^T (dP^T - Di), branchless mask
      const int qg = q0w + col;
      const __bf16* brow = nullptr;
      if (BIASED)
        brow = bias + bias_base + (long)min(qg, sq - 1) * skv;
      const bool need_mask =
          (causal && kv0 + KB > q0w + off + 1) || (kv0 + KB > skv) ||
          (WINDOWED && kv0 < q0w + 32 + off - window + 1);
      float dsv[16];
      float bv[16];
      if (BIASED)
        load_bias16(brow, kv0 + 4 * hi, skv, bv);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sc = st[r] * sl2e;
        if (BIASED) sc += bv[r];
        const float e = __expf(sc - lse_c);
        float pr = e;
        if (need_mask) {
          const int key = kv0 + mfma32_d_row(lane, r);
          const bool ok = key < skv && !(causal && key > qg + off) &&
                          !(WINDOWED && key <= qg + off - window);
          pr = ok ? e : 0.f;
        }
        dsv[r] = pr * (dp[r] - di_c) * scale;
      }

      // pack dS^T to MFMA B-fragments (cvt_pk + permlane32_swap)
      unsigned w[8];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int g = 0; g < 2; ++g) {
          unsigned a = pack_bf16(dsv[8 * i + 2 * g], dsv[8 * i + 2 * g + 1]);
          unsigned bb = pack_bf16(dsv[8 * i + 2 * g + 4],
                                  dsv[8 * i + 2 * g + 5]);
          auto r2 = __builtin_amdgcn_permlane32_swap(a, bb, false, false);
          w[i * 4 + g] = r2[0];
          w[i * 4 + 2 + g] = r2[1];
        }
      }

      // dQ^T += K^T dS
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        union { unsigned u[4]; bf16x8 f; } db;
#pragma unroll
        for (int g = 0; g < 4; ++g) db.u[g] = w[kh * 4 + g];
#pragma unroll
        for (int dt = 0; dt < ND; ++dt) {
          bf16x8 ka = *reinterpret_cast<const bf16x8*>(
              kt_lds + (col + 32 * dt) * KTROW + kh * 16 + 8 * hi);
          dq_acc[dt] = mfma32_bf16(ka, db.f, dq_acc[dt]);
        }
      }
    }

    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: dQ^T[d][q=lane]
  const int qg = q0w + col;
  if (qg < sq) {
    __bf16* dqr = dq + q_base + (long)qg * q_stride;
#pragma unroll
    for (int dt = 0; dt < ND; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        dqr[dt * 32 + mfma32_d_row(lane, r)] = (__bf16)dq_acc[dt][r];
  }
}

template <int D, bool BIASED = false, bool WINDOWED = false>
__global__ __launch_bounds__(512, 2)
void flash_bwd_dq_kernel(const __bf16* __restrict__ dout,
                         const __bf16* __restrict__ q,
                         const __bf16* __restrict__ k,
                         const __bf16* __restrict__ v,
                         const float* __restrict__ lse,
                         const float* __restrict__ di,
                         __bf16* __restrict__ dq,
                         int b, int sq, int skv, int hq, int hkv,
                         float scale, bool causal,
                         const __bf16* __restrict__ bias = nullptr,
                         bool paired = true, bool sbhd = false,
                         int window = 0) {
  constexpr int KB = 32, QBF = 256;
  constexpr int BUFSZ = 2 * KB * (D + 8) + D * (KB + 8);
  __shared__ __align__(16) __bf16 smem[2 * BUFSZ];
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);
  const long q_base = sbhd ? ((long)bi * hq + h) * D
                           : ((long)bi * sq * hq + h) * D;
  const long kv_base = sbhd ? ((long)bi * hkv + hk) * D
                            : ((long)bi * skv * hkv + hk) * D;
  const int q_str = (sbhd ? b * hq : hq) * D;
  const int kv_str = (sbhd ? b * hkv : hkv) * D;
  const long lse_base = ((long)bi * hq + h) * sq;
  const long bias_base = (long)h * sq * skv;
  const int off = skv - sq;
  const int nqb = (sq + QBF - 1) / QBF;
  flash_bwd_dq_block<D, BIASED, WINDOWED>(blockIdx.x, dout, q, k, v, lse, di, dq,
                                smem, q_base, kv_base, lse_base, q_str,
                                kv_str, off, sq, skv, scale, causal, bias,
                                bias_base, window);
  const int qb2 = nqb - 1 - (int)blockIdx.x;
  if (causal && paired && qb2 > (int)blockIdx.x) {
    __syncthreads();
    flash_bwd_dq_block<D, BIASED, WINDOWED>(qb2, dout, q, k, v, lse, di, dq, smem,
                                  q_base, kv_base, lse_base, q_str,
                                  kv_str, off, sq, skv, scale, causal,
                                  bias, bias_base, window);
  }
}

// ---------------------------------------------------------------------------
// backward dK/dV (v5): expanded per q-head (host reduces over the GQA
// group), 8 waves x 32 keys (256 keys/workgroup), async-staged q tiles.
//
// v5 = two SEQUENTIAL phases in one kernel (dV pass, then dK pass), each
// with only 64 accumulator VGPRs so the wave's K (and V) fragments are
// PRELOADED into loop-invariant registers.  The v3/v4 single-pass variant
// kept 128 accumulator VGPRs live (dkt+dvt) and the compiler was forced
// to re-load every K/V fragment from L2 one at a time with a full
// vmcnt(0) drain before EACH MFMA — rocprofv3 PMC showed the waves parked
// (SQ_WAIT_ANY 73.7% of wave cycles, MFMA busy 28%) while the healthy dq
// kernel sits at 46%/51%.  The S matrix is computed twice (once per
// phase): +25% MFMA work for register room — measured net win.
//   phase dV:  S = mfma(Q_frag[from q_lds rows], K^T_frag[kfr regs])
//              dV^T[d][key] = mfma(dO^T_frag[from t_lds], P_frag[permlane])
//   phase dK:  S again; dP = mfma(dO_frag[from do_lds rows], V^T[vfr regs])
//              dK^T[d][key] = mfma(Q^T_frag[from t_lds], dS_frag[permlane])
// ---------------------------------------------------------------------------
template <int D, bool DKPH, bool BIASED = false, bool WINDOWED = false>
__device__ __attribute__((noinline))
void flash_bwd_dkv_phase(int kvblk, const __bf16* __restrict__ dout,
                         const __bf16* __restrict__ q,
                         const __bf16* __restrict__ k,
                         const __bf16* __restrict__ v,
                         const float* __restrict__ lse,
                         const float* __restrict__ di,
                         __bf16* __restrict__ dk_exp,
                         __bf16* __restrict__ dv_exp, __bf16* smem_base,
                         float* lsedi_base, long q_base, long kv_base,
                         long dkv_base, long lse_base, int q_stride,
                         int kv_stride, int dkv_stride, int off, int sq,
                         int skv, float scale, bool causal,
                         const __bf16* __restrict__ bias = nullptr,
                         float* __restrict__ dbias = nullptr,
                         long bias_base = 0, int window = 0) {
  constexpr int QT = 64;           // q tile (2 mfma halves per stage)
  constexpr int KBW = 256;         // keys per workgroup (8 waves x 32)
  constexpr int QROW = D + 8;
  constexpr int TROW = QT + 8;
  constexpr int NK = D / 16;
  constexpr int ND = D / 32;
  constexpr int QPT = (QT * D / 8 + 511) / 512;   // row packs per thread

  // LDS per buffer: q rows + transposed image (+ dO rows in the dK phase)
  constexpr int BUFSZ = (DKPH ? 2 : 1) * QT * QROW + D * TROW;
  __bf16* smem = smem_base;
  float (*lse_lds)[QT] = reinterpret_cast<float (*)[QT]>(lsedi_base);
  float (*di_lds)[QT] = reinterpret_cast<float (*)[QT]>(lsedi_base + 2 * QT);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;   // this wave's key index
  const int hi = lane >> 5;
  const int k0w = kvblk * KBW + wid * 32;  // this wave's first key row

  // loop-invariant K (and V) fragments, preloaded into registers — the
  // whole point of the phase split
  const int kg_f = min(k0w + col, skv - 1);
  bf16x8 kfr[NK];
  bf16x8 vfr[DKPH ? NK : 1];
  {
    const __bf16* kp_f = k + kv_base + (long)kg_f * kv_stride + 8 * hi;
    const __bf16* vp_f = v + kv_base + (long)kg_f * kv_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      kfr[ks] = *reinterpret_cast<const bf16x8*>(kp_f + ks * 16);
      if (DKPH)
        vfr[ks] = *reinterpret_cast<const bf16x8*>(vp_f + ks * 16);
    }
  }

  f32x16 acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) acc[dt] = (f32x16)(0.f);

  // first q tile that can attend any key of this block (per-wave causal
  // skipping happens per 32-q half inside the loop); with a sliding
  // window, also the LAST q that can see any of this block's keys:
  // q <= key - off + window - 1
  int qstart = 0;
  if (causal) qstart = max(0, ((kvblk * KBW - off) / QT) * QT);
  int qstop = sq;
  if (WINDOWED)
    qstop = min(sq, kvblk * KBW + KBW - 1 - off + window);

  // staged registers: q rows, transposed image of (dV: dO / dK: Q),
  // dO rows (dK phase only)
  bf16x8 qst[QPT], dost[QPT];
  ushort8 ttst[QPT];
  float lse_st, di_st;
  const __bf16* tsrc = DKPH ? q : dout;  // transposed-image source

  auto stage_load = [&](int qt0) {
    const bool full = (qt0 + QT <= sq);
#pragma unroll
    for (int p = 0; p < QPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= QT * D / 8) break;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      const int qg = full ? qt0 + row : min(qt0 + row, sq - 1);
      qst[p] = *reinterpret_cast<const bf16x8*>(
          q + q_base + (long)qg * q_stride + c8);
      if (DKPH)
        dost[p] = *reinterpret_cast<const bf16x8*>(
            dout + q_base + (long)qg * q_stride + c8);
      if (!full && qt0 + row >= sq) {
        qst[p] = (bf16x8)(__bf16(0.f));
        if (DKPH) dost[p] = (bf16x8)(__bf16(0.f));
      }
    }
#pragma unroll
    for (int p = 0; p < QPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= D * QT / 8) break;
      const int c = idx & (D - 1);
      const int qc = (idx / D) * 8;
      const unsigned short* tb = reinterpret_cast<const unsigned short*>(
          tsrc + q_base) + c;
      ushort8 t8;
      if (full) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          t8[j] = tb[(long)(qt0 + qc + j) * q_stride];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned short a = tb[(long)min(qt0 + qc + j, sq - 1) * q_stride];
          t8[j] = qt0 + qc + j < sq ? a : (unsigned short)0;
        }
      }
      ttst[p] = t8;
    }
    if (tid < QT) {
      const int qg = qt0 + tid;
      lse_st = qg < sq ? lse[lse_base + qg] : INFINITY;
      if (DKPH) di_st = qg < sq ? di[lse_base + qg] : 0.f;
    }
  };
  auto stage_write = [&](int buf) {
    __bf16* q_lds = smem + buf * BUFSZ;
    __bf16* t_lds = q_lds + QT * QROW;
    __bf16* do_lds = t_lds + D * TROW;
#pragma unroll
    for (int p = 0; p < QPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= QT * D / 8) break;
      const int row = idx / (D / 8);
      const int c8 = (idx - row * (D / 8)) * 8;
      *reinterpret_cast<bf16x8*>(q_lds + row * QROW + c8) = qst[p];
      if (DKPH)
        *reinterpret_cast<bf16x8*>(do_lds + row * QROW + c8) = dost[p];
    }
#pragma unroll
    for (int p = 0; p < QPT; ++p) {
      const int idx = tid + p * 512;
      if (idx >= D * QT / 8) break;
      const int c = idx & (D - 1);
      const int qc = (idx / D) * 8;
      *reinterpret_cast<ushort8*>(t_lds + c * TROW + qc) = ttst[p];
    }
    if (tid < QT) {
      lse_lds[buf][tid] = lse_st;
      if (DKPH) di_lds[buf][tid] = di_st;
    }
  };

  stage_load(qstart);
  stage_write(0);
  __syncthreads();
  int cur = 0;

  for (int qt0 = qstart; qt0 < qstop; qt0 += QT) {
    const bool have_next = qt0 + QT < qstop;
    const __bf16* q_lds = smem + cur * BUFSZ;
    const __bf16* t_lds = q_lds + QT * QROW;
    const __bf16* do_lds = t_lds + D * TROW;
    if (have_next) stage_load(qt0 + QT);

    // two 32-q mfma halves per staged tile
#pragma unroll
    for (int qh = 0; qh < QT / 32; ++qh) {
      const int q0h = qt0 + qh * 32;
      // per-half skips (wave-uniform): any key of this wave live?
      if (causal && q0h + 31 + off < k0w) continue;
      if (WINDOWED && q0h + off - window + 1 > k0w + 31) continue;
      if (q0h >= sq) continue;

      // S = Q K^T (; dP = dO V^T)  D-layout rows=q(crow), cols=key(lane);
      // A-frag lane l&31 = q row of the half, B-frag lane = key
      f32x16 s_acc = (f32x16)(0.f), dp_acc = (f32x16)(0.f);
#pragma unroll
      for (int ks = 0; ks < NK; ++ks) {
        bf16x8 qa = *reinterpret_cast<const bf16x8*>(
            q_lds + (qh * 32 + col) * QROW + ks * 16 + 8 * hi);
        s_acc = mfma32_bf16(qa, kfr[ks], s_acc);
        if (DKPH) {
          bf16x8 doa = *reinterpret_cast<const bf16x8*>(
              do_lds + (qh * 32 + col) * QROW + ks * 16 + 8 * hi);
          dp_acc = mfma32_bf16(doa, vfr[ks], dp_acc);
        }
      }

      float wv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = mfma32_d_row(lane, r);
        const float lse_r = lse_lds[cur][qh * 32 + qrow];
        float sc = s_acc[r] * scale;
        if (BIASED) {
          const long bq = (long)min(q0h + qrow, sq - 1);
          sc += (float)bias[bias_base + bq * skv +
                            min(k0w + col, skv - 1)];
        }
        const float e = __expf(sc - lse_r);
        float p = e;
        if (WINDOWED) {
          const int keyg = k0w + col;
          const int qo = q0h + qrow + off;
          const bool ok = keyg < skv && (!causal || keyg <= qo) &&
                          keyg > qo - window;
          p = ok ? e : 0.f;
        } else if (causal) {
          const int keyg = k0w + col;
          const bool ok = keyg < skv && keyg <= q0h + qrow + off;
          p = ok ? e : 0.f;
        } else if (k0w + col >= skv) {
          p = 0.f;
        }
        if (DKPH) {
          const float di_r = di_lds[cur][qh * 32 + qrow];
          const float t = p * (dp_acc[r] - di_r);
          if (BIASED) {
            // d(bias) = dS (un-scaled); batch contributions accumulate
            // via fp32 atomics into [hq, sq, skv]
            const int bq = q0h + qrow, bk = k0w + col;
            if (bq < sq && bk < skv)
              atomicAdd(dbias + bias_base + (long)bq * skv + bk, t);
          }
          wv[r] = t * scale;
        } else {
          wv[r] = p;
        }
      }

      // convert P / dS (D-layout rows=q) to B-fragments (lane=key,
      // in-lane=q) via cvt_pk + permlane32_swap
      unsigned w[8];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int g = 0; g < 2; ++g) {
          unsigned a = pack_bf16(wv[8 * i + 2 * g], wv[8 * i + 2 * g + 1]);
          unsigned bb = pack_bf16(wv[8 * i + 2 * g + 4],
                                  wv[8 * i + 2 * g + 5]);
          auto r2 = __builtin_amdgcn_permlane32_swap(a, bb, false, false);
          w[i * 4 + g] = r2[0];
          w[i * 4 + 2 + g] = r2[1];
        }
      }

      // dV^T += dO^T P  /  dK^T += Q^T dS  (two K=16 slices of the half)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        union { unsigned u[4]; bf16x8 f; } fb;
#pragma unroll
        for (int g = 0; g < 4; ++g) fb.u[g] = w[kh * 4 + g];
#pragma unroll
        for (int dt = 0; dt < ND; ++dt) {
          bf16x8 ta = *reinterpret_cast<const bf16x8*>(
              t_lds + (col + 32 * dt) * TROW + qh * 32 + kh * 16 + 8 * hi);
          acc[dt] = mfma32_bf16(ta, fb.f, acc[dt]);
        }
      }
    }

    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: dK^T/dV^T D-layout rows=d(crow), cols=key(lane)
  const int keyg = k0w + col;
  if (keyg < skv) {
    __bf16* outr = (DKPH ? dk_exp : dv_exp) + dkv_base +
                   (long)keyg * dkv_stride;
#pragma unroll
    for (int dt = 0; dt < ND; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        outr[dt * 32 + mfma32_d_row(lane, r)] = (__bf16)acc[dt][r];
  }
}

template <int D, bool BIASED = false, bool WINDOWED = false>
__device__ void flash_bwd_dkv_block(
    int kvblk, const __bf16* __restrict__ dout, const __bf16* __restrict__ q,
    const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ di,
    __bf16* __restrict__ dk_exp, __bf16* __restrict__ dv_exp,
    __bf16* smem_base, float* lsedi_base, long q_base, long kv_base,
    long dkv_base, long lse_base, int q_stride, int kv_stride,
    int dkv_stride, int off, int sq, int skv, float scale, bool causal,
    const __bf16* bias = nullptr, float* dbias = nullptr,
    long bias_base = 0, int window = 0) {
  flash_bwd_dkv_phase<D, false, BIASED, WINDOWED>(
      kvblk, dout, q, k, v, lse, di, dk_exp, dv_exp, smem_base, lsedi_base,
      q_base, kv_base, dkv_base, lse_base, q_stride, kv_stride, dkv_stride,
      off, sq, skv, scale, causal, bias, dbias, bias_base, window);
  __syncthreads();
  flash_bwd_dkv_phase<D, true, BIASED, WINDOWED>(
      kvblk, dout, q, k, v, lse, di, dk_exp, dv_exp, smem_base, lsedi_base,
      q_base, kv_base, dkv_base, lse_base, q_stride, kv_stride, dkv_stride,
      off, sq, skv, scale, causal, bias, dbias, bias_base, window);
}

template <int D, bool BIASED = false, bool WINDOWED = false>
__global__ __launch_bounds__(512, 2)
void flash_bwd_dkv_kernel(const __bf16* __restrict__ dout,
                          const __bf16* __restrict__ q,
                          const __bf16* __restrict__ k,
                          const __bf16* __restrict__ v,
                          const float* __restrict__ lse,
                          const float* __restrict__ di,
                          __bf16* __restrict__ dk_exp,   // [b, skv, hq, D]
                          __bf16* __restrict__ dv_exp,
                          int b, int sq, int skv, int hq, int hkv,
                          float scale, bool causal,
                          const __bf16* __restrict__ bias = nullptr,
                          float* __restrict__ dbias = nullptr,
                          bool paired = true, bool sbhd = false,
                          int window = 0) {
  constexpr int QT = 64, KBW = 256;  // must match flash_bwd_dkv_phase
  // dK phase is the larger LDS user: q rows + dO rows + transposed image
  constexpr int BUFSZ = 2 * QT * (D + 8) + D * (QT + 8);
  __shared__ __align__(16) __bf16 smem[2 * BUFSZ];
  __shared__ __align__(16) float lsedi[4 * QT];
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);
  const long q_base = sbhd ? ((long)bi * hq + h) * D
                           : ((long)bi * sq * hq + h) * D;
  const long kv_base = sbhd ? ((long)bi * hkv + hk) * D
                            : ((long)bi * skv * hkv + hk) * D;
  // dk_exp/dv_exp are [b, skv, hq, D] (bshd) or [skv, b, hq, D] (sbhd)
  const long dkv_base = sbhd ? ((long)bi * hq + h) * D
                             : ((long)bi * skv * hq + h) * D;
  const int q_str = (sbhd ? b * hq : hq) * D;
  const int kv_str = (sbhd ? b * hkv : hkv) * D;
  const int dkv_str = (sbhd ? b * hq : hq) * D;
  const long lse_base = ((long)bi * hq + h) * sq;
  const int off = skv - sq;
  const int nkb = (skv + KBW - 1) / KBW;
  const long bias_base = (long)h * sq * skv;
  flash_bwd_dkv_block<D, BIASED, WINDOWED>(blockIdx.x, dout, q, k, v, lse, di, dk_exp,
                                 dv_exp, smem, lsedi, q_base, kv_base,
                                 dkv_base, lse_base, q_str, kv_str,
                                 dkv_str, off, sq, skv, scale, causal, bias,
                                 dbias, bias_base, window);
  const int kb2 = nkb - 1 - (int)blockIdx.x;
  if (causal && paired && kb2 > (int)blockIdx.x) {
    __syncthreads();
    flash_bwd_dkv_block<D, BIASED, WINDOWED>(kb2, dout, q, k, v, lse, di, dk_exp,
                                   dv_exp, smem, lsedi, q_base, kv_base,
                                   dkv_base, lse_base, q_str, kv_str,
                                   dkv_str, off, sq, skv, scale, causal,
                                   bias, dbias, bias_base, window);
  }
}

// ---------------------------------------------------------------------------
// MFMA layout probe: one-wave 32x32x16 GEMM from global memory laid out by
// the assumed fragment maps. Validated against torch matmul on-device.
// ---------------------------------------------------------------------------
template <bool ALT>
__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,  // [32][16]
                                  const __bf16* __restrict__ B,  // [16][32]
                                  float* __restrict__ Dst) {     // [32][32]
  const int lane = threadIdx.x & 63;
  bf16x8 a, bfrag;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int kk = ALT ? mfma32_ab_k_alt(lane, j) : mfma32_ab_k(lane, j);
    a[j] = A[(lane & 31) * 16 + kk];
    bfrag[j] = B[kk * 32 + (lane & 31)];
  }
  f32x16 d = mfma32_bf16(a, bfrag, (f32x16)(0.f));
#pragma unroll
  for (int i = 0; i < 16; ++i)
    Dst[mfma32_d_row(lane, i) * 32 + mfma32_d_col(lane)] = d[i];
}

}  // namespace

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------
template <int D>
static void flash_fwd_launch_d(const __bf16* q, const __bf16* k,
                               const __bf16* v, __bf16* o, float* lse, int b,
                               int sq, int skv, int hq, int hkv, float scale,
                               bool causal, const __bf16* bias,
                               bool sbhd, int window, hipStream_t st) {
  int nqb = (sq + 255) / 256;
  // complementary-pair causal scheduling halves the grid; skip it when
  // the halved grid underfills the 256-CU chip (small-seq shapes)
  const bool paired = causal && ((nqb + 1) / 2) * (long)b * hq >= 256;
  dim3 grid((causal && paired) ? (nqb + 1) / 2 : nqb, b * hq);
  if (bias != nullptr)
    hipLaunchKernelGGL((flash_fwd_kernel<D, true>), grid, dim3(512), 0, st,
                       q, k, v, o, lse, b, sq, skv, hq, hkv, scale, causal,
                       bias, paired, sbhd, window);
  else if (window > 0)
    hipLaunchKernelGGL((flash_fwd_kernel<D, false, true>), grid, dim3(512),
                       0, st, q, k, v, o, lse, b, sq, skv, hq, hkv, scale,
                       causal, nullptr, paired, sbhd, window);
  else
    hipLaunchKernelGGL((flash_fwd_kernel<D>), grid, dim3(512), 0, st, q, k,
                       v, o, lse, b, sq, skv, hq, hkv, scale, causal,
                       nullptr, paired, sbhd, 0);
}

void flash_fwd_launch(const __bf16* q, const __bf16* k, const __bf16* v,
                      __bf16* o, float* lse, int b, int sq, int skv, int hq,
                      int hkv, int d, float scale, bool causal,
                      hipStream_t st, const __bf16* bias, bool sbhd,
                      int window) {
  if (d == 64)
    flash_fwd_launch_d<64>(q, k, v, o, lse, b, sq, skv, hq, hkv, scale,
                           causal, bias, sbhd, window, st);
  else
    flash_fwd_launch_d<128>(q, k, v, o, lse, b, sq, skv, hq, hkv, scale,
                            causal, bias, sbhd, window, st);
}

void attn_di_launch(const __bf16* dout, const __bf16* o, float* di, int b,
                    int sq, int hq, int d, hipStream_t st,
                    bool sbhd = false) {
  long rows = (long)b * sq * hq;
  int grid = galv_grid((rows + 3) / 4);
  if (d == 64)
    hipLaunchKernelGGL((attn_di_kernel<64>), dim3(grid), dim3(256), 0, st,
                       dout, o, di, b, sq, hq, sbhd);
  else
    hipLaunchKernelGGL((attn_di_kernel<128>), dim3(grid), dim3(256), 0, st,
                       dout, o, di, b, sq, hq, sbhd);
}

template <int D>
static void flash_bwd_launch_d(const __bf16* dout, const __bf16* q,
                               const __bf16* k, const __bf16* v,
                               const float* lse, const float* di, __bf16* dq,
                               __bf16* dk_exp, __bf16* dv_exp, int b, int sq,
                               int skv, int hq, int hkv, float scale,
                               bool causal, const __bf16* bias, float* dbias,
                               bool sbhd, int window, hipStream_t st) {
  int nqb = (sq + 255) / 256;
  const bool pq = causal && ((nqb + 1) / 2) * (long)b * hq >= 256;
  dim3 gq((causal && pq) ? (nqb + 1) / 2 : nqb, b * hq);
  int nkb = (skv + 255) / 256;
  const bool pkv = causal && ((nkb + 1) / 2) * (long)b * hq >= 256;
  dim3 gkv((causal && pkv) ? (nkb + 1) / 2 : nkb, b * hq);
  if (bias != nullptr) {
    hipLaunchKernelGGL((flash_bwd_dq_kernel<D, true>), gq, dim3(512), 0, st,
                       dout, q, k, v, lse, di, dq, b, sq, skv, hq, hkv,
                       scale, causal, bias, pq, sbhd, window);
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<D, true>), gkv, dim3(512), 0,
                       st, dout, q, k, v, lse, di, dk_exp, dv_exp, b, sq,
                       skv, hq, hkv, scale, causal, bias, dbias, pkv, sbhd,
                       window);
  } else if (window > 0) {
    hipLaunchKernelGGL((flash_bwd_dq_kernel<D, false, true>), gq, dim3(512),
                       0, st, dout, q, k, v, lse, di, dq, b, sq, skv, hq,
                       hkv, scale, causal, nullptr, pq, sbhd, window);
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<D, false, true>), gkv,
                       dim3(512), 0, st, dout, q, k, v, lse, di, dk_exp,
                       dv_exp, b, sq, skv, hq, hkv, scale, causal, nullptr,
                       nullptr, pkv, sbhd, window);
  } else {
    hipLaunchKernelGGL((flash_bwd_dq_kernel<D>), gq, dim3(512), 0, st, dout,
                       q, k, v, lse, di, dq, b, sq, skv, hq, hkv, scale,
                       causal, nullptr, pq, sbhd, 0);
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<D>), gkv, dim3(512), 0, st,
                       dout, q, k, v, lse, di, dk_exp, dv_exp, b, sq, skv,
                       hq, hkv, scale, causal, nullptr, nullptr, pkv, sbhd,
                       0);
  }
}

void flash_bwd_launch(const __bf16* dout, const __bf16* q, const __bf16* k,
                      const __bf16* v, const float* lse, const float* di,
                      __bf16* dq, __bf16* dk_exp, __bf16* dv_exp, int b,
                      int sq, int skv, int hq, int hkv, int d, float scale,
                      bool causal, hipStream_t st, const __bf16* bias,
                      float* dbias, bool sbhd, int window) {
  if (d == 64)
    flash_bwd_launch_d<64>(dout, q, k, v, lse, di, dq, dk_exp, dv_exp, b, sq,
                           skv, hq, hkv, scale, causal, bias, dbias, sbhd,
                           window, st);
  else
    flash_bwd_launch_d<128>(dout, q, k, v, lse, di, dq, dk_exp, dv_exp, b,
                            sq, skv, hq, hkv, scale, causal, bias, dbias,
                            sbhd, window, st);
}

void mfma_probe_launch(const __bf16* A, const __bf16* B, float* Dst, bool alt,
                       hipStream_t st) {
  if (alt)
    hipLaunchKernelGGL((mfma_probe_kernel<true>), dim3(1), dim3(64), 0, st,
                       A, B, Dst);
  else
    hipLaunchKernelGGL((mfma_probe_kernel<false>), dim3(1), dim3(64), 0, st,
                       A, B, Dst);
}
