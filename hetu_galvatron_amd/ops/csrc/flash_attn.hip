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
// v1 structure (correctness-first, MFMA throughout):
//   forward: 4-wave workgroup = 128 q rows (32/wave), KV tiles of 32 staged
//   in LDS (K row-major padded, V transposed), online softmax with the
//   D-layout row map of v_mfma_f32_32x32x16_bf16, P staged through a
//   per-wave LDS tile to re-enter as the PV A-operand.
//   backward: separate dq and dkv kernels (no atomics); dkv computes
//   per-q-head expanded dK/dV, host sums over the GQA group.
// The MFMA A/B lane maps assumed here are validated on-device by
// galv_mfma_probe (tests/ops/test_gpu_kernels.py::test_mfma_layout).
#include "common.h"

namespace {

constexpr int QB = 128;   // q rows per workgroup (4 waves x 32)
constexpr int KVB = 32;   // kv tile

DEV_INLINE float neg_big() { return -1e30f; }

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 2)
void flash_fwd_kernel(const __bf16* __restrict__ q,
                      const __bf16* __restrict__ k,
                      const __bf16* __restrict__ v,
                      __bf16* __restrict__ o, float* __restrict__ lse,
                      int b, int sq, int skv, int hq, int hkv, float scale,
                      bool causal) {
  constexpr int KROW = D + 8;        // padded K-tile row (bf16 elems)
  constexpr int VTROW = KVB + 8;     // padded transposed-V row
  constexpr int PROW = KVB + 8;      // per-wave P staging row
  constexpr int NK = D / 16;         // K-slices per S tile
  constexpr int ND = D / 32;         // 32-wide d tiles of O

  __shared__ __align__(16) __bf16 smem[KVB * KROW + D * VTROW + 4 * KVB * PROW];
  __bf16* k_lds = smem;
  __bf16* vt_lds = smem + KVB * KROW;
  __bf16* p_lds = vt_lds + D * VTROW;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;   // MFMA col (key index / d index)
  const int hi = lane >> 5;

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);

  const long q_base = ((long)bi * sq * hq + h) * D;
  const long kv_base_k = ((long)bi * skv * hkv + hk) * D;
  const int q_stride = hq * D;
  const int kv_stride = hkv * D;
  const int off = skv - sq;

  const int q0w = qblk * QB + wid * KVB;  // this wave's first q row
  // A-fragments of Q: lane holds Q[q0w+col][ks*16 + 8*hi + j]
  bf16x8 qf[NK];
  {
    const int qg = min(q0w + col, sq - 1);
    const __bf16* qp = q + q_base + (long)qg * q_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks)
      qf[ks] = *reinterpret_cast<const bf16x8*>(qp + ks * 16);
  }

  f32x16 ov[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) ov[dt] = (f32x16)(0.f);
  float m_run[16], l_run[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) { m_run[i] = neg_big(); l_run[i] = 0.f; }

  int kv_end = skv;
  if (causal) kv_end = min(skv, qblk * QB + QB + off);
  __bf16* my_p = p_lds + wid * KVB * PROW;

  for (int kv0 = 0; kv0 < kv_end; kv0 += KVB) {
    // ---- stage K (row-major, padded) and V (transposed) ----
    __syncthreads();
    constexpr int PACKS = KVB * D / 8;  // 16B packs in the tile
#pragma unroll
    for (int p = tid; p < PACKS; p += 256) {
      const int row = p / (D / 8);
      const int c8 = (p - row * (D / 8)) * 8;
      const int kg = kv0 + row;
      bf16x8 kvv = (bf16x8)(__bf16(0.f));
      if (kg < skv)
        kvv = *reinterpret_cast<const bf16x8*>(
            k + kv_base_k + (long)kg * kv_stride + c8);
      *reinterpret_cast<bf16x8*>(k_lds + row * KROW + c8) = kvv;
      bf16x8 vv = (bf16x8)(__bf16(0.f));
      if (kg < skv)
        vv = *reinterpret_cast<const bf16x8*>(
            v + kv_base_k + (long)kg * kv_stride + c8);
#pragma unroll
      for (int i = 0; i < 8; ++i) vt_lds[(c8 + i) * VTROW + row] = vv[i];
    }
    __syncthreads();

    // ---- S = Q K^T ----
    f32x16 s_acc = (f32x16)(0.f);
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      bf16x8 kb = *reinterpret_cast<const bf16x8*>(
          k_lds + col * KROW + ks * 16 + 8 * hi);
      s_acc = mfma32_bf16(qf[ks], kb, s_acc);
    }

    // ---- mask + online softmax ----
    float sv[16];
    const int colg = kv0 + col;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const int rowg = q0w + mfma32_d_row(lane, i);
      float x = s_acc[i] * scale;
      if (colg >= skv || (causal && colg > rowg + off)) x = neg_big();
      sv[i] = x;
    }
    float alpha[16];
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const float tm = half_wave_max(sv[i]);
      const float mn = fmaxf(m_run[i], tm);
      alpha[i] = (m_run[i] <= neg_big()) ? 0.f : __expf(m_run[i] - mn);
      m_run[i] = mn;
    }
    float pv[16];
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      pv[i] = (m_run[i] <= neg_big()) ? 0.f : __expf(sv[i] - m_run[i]);
      l_run[i] = l_run[i] * alpha[i] + half_wave_sum(pv[i]);
      my_p[mfma32_d_row(lane, i) * PROW + col] = (__bf16)pv[i];
    }
#pragma unroll
    for (int dt = 0; dt < ND; ++dt)
#pragma unroll
      for (int i = 0; i < 16; ++i) ov[dt][i] *= alpha[i];

    // ---- O += P V ----  (P re-read as A-fragments from this wave's LDS)
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          my_p + col * PROW + kh * 16 + 8 * hi);
#pragma unroll
      for (int dt = 0; dt < ND; ++dt) {
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            vt_lds + (dt * 32 + col) * VTROW + kh * 16 + 8 * hi);
        ov[dt] = mfma32_bf16(pa, vb, ov[dt]);
      }
    }
  }

  // ---- epilogue ----
  float inv_l[16];
#pragma unroll
  for (int i = 0; i < 16; ++i)
    inv_l[i] = l_run[i] > 0.f ? 1.f / l_run[i] : 0.f;
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int rowg = q0w + mfma32_d_row(lane, i);
    if (rowg < sq) {
      __bf16* orow = o + q_base + (long)rowg * q_stride;
#pragma unroll
      for (int dt = 0; dt < ND; ++dt)
        orow[dt * 32 + col] = (__bf16)(ov[dt][i] * inv_l[i]);
      if (col == 0)
        lse[((long)bi * hq + h) * sq + rowg] =
            l_run[i] > 0.f ? m_run[i] + __logf(l_run[i])
                           : -INFINITY;
    }
  }
}

// ---------------------------------------------------------------------------
// Di = rowsum(dO * O)  — one wave per (b, s, h) row
// ---------------------------------------------------------------------------
template <int D>
__global__ void attn_di_kernel(const __bf16* __restrict__ dout,
                               const __bf16* __restrict__ o,
                               float* __restrict__ di, int b, int sq, int hq) {
  const long rows = (long)b * sq * hq;  // row index = (bi*sq + s)*hq + h
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
      const long bi = row / ((long)sq * hq);
      const long rem = row - bi * sq * hq;
      const long s = rem / hq;
      const long h = rem - s * hq;
      di[(bi * hq + h) * sq + s] = acc;
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ: grid over q blocks; loops kv tiles
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 2)
void flash_bwd_dq_kernel(const __bf16* __restrict__ dout,
                         const __bf16* __restrict__ q,
                         const __bf16* __restrict__ k,
                         const __bf16* __restrict__ v,
                         const float* __restrict__ lse,
                         const float* __restrict__ di,
                         __bf16* __restrict__ dq,
                         int b, int sq, int skv, int hq, int hkv,
                         float scale, bool causal) {
  constexpr int KROW = D + 8;
  constexpr int KTROW = KVB + 8;
  constexpr int PROW = KVB + 8;
  constexpr int NK = D / 16;
  constexpr int ND = D / 32;

  __shared__ __align__(16) __bf16
      smem[KVB * KROW /*K*/ + D * KTROW /*KT*/ + KVB * KROW /*V*/ +
           4 * KVB * PROW /*dS per wave*/];
  __bf16* k_lds = smem;
  __bf16* kt_lds = smem + KVB * KROW;
  __bf16* v_lds = kt_lds + D * KTROW;
  __bf16* ds_lds = v_lds + KVB * KROW;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);

  const long q_base = ((long)bi * sq * hq + h) * D;
  const long kv_base = ((long)bi * skv * hkv + hk) * D;
  const int q_stride = hq * D;
  const int kv_stride = hkv * D;
  const int off = skv - sq;
  const int q0w = qblk * QB + wid * KVB;

  bf16x8 qf[NK], dof[NK];
  {
    const int qg = min(q0w + col, sq - 1);
    const __bf16* qp = q + q_base + (long)qg * q_stride + 8 * hi;
    const __bf16* dop = dout + q_base + (long)qg * q_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      qf[ks] = *reinterpret_cast<const bf16x8*>(qp + ks * 16);
      dof[ks] = *reinterpret_cast<const bf16x8*>(dop + ks * 16);
    }
  }
  float lse_r[16], di_r[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int rowg = q0w + mfma32_d_row(lane, i);
    if (rowg < sq) {
      lse_r[i] = lse[((long)bi * hq + h) * sq + rowg];
      di_r[i] = di[((long)bi * hq + h) * sq + rowg];
    } else {
      lse_r[i] = INFINITY;  // exp(x - inf) = 0 -> junk rows contribute 0
      di_r[i] = 0.f;
    }
  }

  f32x16 dq_acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) dq_acc[dt] = (f32x16)(0.f);

  int kv_end = skv;
  if (causal) kv_end = min(skv, qblk * QB + QB + off);
  __bf16* my_ds = ds_lds + wid * KVB * PROW;

  for (int kv0 = 0; kv0 < kv_end; kv0 += KVB) {
    __syncthreads();
    constexpr int PACKS = KVB * D / 8;
#pragma unroll
    for (int p = tid; p < PACKS; p += 256) {
      const int row = p / (D / 8);
      const int c8 = (p - row * (D / 8)) * 8;
      const int kg = kv0 + row;
      bf16x8 kvv = (bf16x8)(__bf16(0.f));
      if (kg < skv)
        kvv = *reinterpret_cast<const bf16x8*>(
            k + kv_base + (long)kg * kv_stride + c8);
      *reinterpret_cast<bf16x8*>(k_lds + row * KROW + c8) = kvv;
#pragma unroll
      for (int i = 0; i < 8; ++i) kt_lds[(c8 + i) * KTROW + row] = kvv[i];
      bf16x8 vv = (bf16x8)(__bf16(0.f));
      if (kg < skv)
        vv = *reinterpret_cast<const bf16x8*>(
            v + kv_base + (long)kg * kv_stride + c8);
      *reinterpret_cast<bf16x8*>(v_lds + row * KROW + c8) = vv;
    }
    __syncthreads();

    // S and dP
    f32x16 s_acc = (f32x16)(0.f), dp_acc = (f32x16)(0.f);
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      bf16x8 kb = *reinterpret_cast<const bf16x8*>(
          k_lds + col * KROW + ks * 16 + 8 * hi);
      s_acc = mfma32_bf16(qf[ks], kb, s_acc);
      bf16x8 vb = *reinterpret_cast<const bf16x8*>(
          v_lds + col * KROW + ks * 16 + 8 * hi);
      dp_acc = mfma32_bf16(dof[ks], vb, dp_acc);
    }

    const int colg = kv0 + col;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const int rowg = q0w + mfma32_d_row(lane, i);
      float p = 0.f;
      if (colg < skv && !(causal && colg > rowg + off))
        p = __expf(s_acc[i] * scale - lse_r[i]);
      const float ds = p * (dp_acc[i] - di_r[i]) * scale;
      my_ds[mfma32_d_row(lane, i) * PROW + col] = (__bf16)ds;
    }

    // dQ += dS @ K   (dS via LDS as A-fragments; K^T via transposed tile)
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8 da = *reinterpret_cast<const bf16x8*>(
          my_ds + col * PROW + kh * 16 + 8 * hi);
#pragma unroll
      for (int dt = 0; dt < ND; ++dt) {
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(
            kt_lds + (dt * 32 + col) * KTROW + kh * 16 + 8 * hi);
        dq_acc[dt] = mfma32_bf16(da, kb, dq_acc[dt]);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int rowg = q0w + mfma32_d_row(lane, i);
    if (rowg < sq) {
      __bf16* dqr = dq + q_base + (long)rowg * q_stride;
#pragma unroll
      for (int dt = 0; dt < ND; ++dt)
        dqr[dt * 32 + col] = (__bf16)dq_acc[dt][i];
    }
  }
}

// ---------------------------------------------------------------------------
// backward dK/dV (expanded per q-head; host reduces over the GQA group).
// grid over kv blocks; loops q tiles.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256, 1)
void flash_bwd_dkv_kernel(const __bf16* __restrict__ dout,
                          const __bf16* __restrict__ q,
                          const __bf16* __restrict__ k,
                          const __bf16* __restrict__ v,
                          const float* __restrict__ lse,
                          const float* __restrict__ di,
                          __bf16* __restrict__ dk_exp,   // [b, skv, hq, D]
                          __bf16* __restrict__ dv_exp,
                          int b, int sq, int skv, int hq, int hkv,
                          float scale, bool causal) {
  constexpr int QROW = D + 8;
  constexpr int TROW = KVB + 8;
  constexpr int PROW = KVB + 8;
  constexpr int NK = D / 16;
  constexpr int ND = D / 32;

  __shared__ __align__(16) __bf16
      smem[KVB * QROW /*Q*/ + D * TROW /*QT*/ + KVB * QROW /*dO*/ +
           D * TROW /*dOT*/ + 4 * KVB * PROW /*P/dS per wave*/];
  __bf16* q_lds = smem;
  __bf16* qt_lds = smem + KVB * QROW;
  __bf16* do_lds = qt_lds + D * TROW;
  __bf16* dot_lds = do_lds + KVB * QROW;
  __bf16* p_lds = dot_lds + D * TROW;
  __shared__ float lse_lds[KVB];
  __shared__ float di_lds[KVB];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;   // q column in S^T
  const int hi = lane >> 5;

  const int kvblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int bi = bh / hq;
  const int h = bh - bi * hq;
  const int hk = h / (hq / hkv);

  const long q_base = ((long)bi * sq * hq + h) * D;
  const long kv_base = ((long)bi * skv * hkv + hk) * D;
  const long dkv_base = ((long)bi * skv * hq + h) * D;
  const int q_stride = hq * D;
  const int kv_stride = hkv * D;
  const int dkv_stride = hq * D;
  const int off = skv - sq;
  const int k0w = kvblk * QB + wid * KVB;  // this wave's first key row

  // K/V A-fragments for this wave's 32 keys
  bf16x8 kf[NK], vf[NK];
  {
    const int kg = min(k0w + col, skv - 1);
    const __bf16* kp = k + kv_base + (long)kg * kv_stride + 8 * hi;
    const __bf16* vp = v + kv_base + (long)kg * kv_stride + 8 * hi;
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      kf[ks] = *reinterpret_cast<const bf16x8*>(kp + ks * 16);
      vf[ks] = *reinterpret_cast<const bf16x8*>(vp + ks * 16);
    }
  }

  f32x16 dk_acc[ND], dv_acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) {
    dk_acc[dt] = (f32x16)(0.f);
    dv_acc[dt] = (f32x16)(0.f);
  }

  // first q row that can attend any key in this block (causal)
  int qstart = 0;
  if (causal) qstart = max(0, ((kvblk * QB - off) / KVB) * KVB);
  __bf16* my_p = p_lds + wid * KVB * PROW;

  for (int qt0 = qstart; qt0 < sq; qt0 += KVB) {
    __syncthreads();
    constexpr int PACKS = KVB * D / 8;
#pragma unroll
    for (int p = tid; p < PACKS; p += 256) {
      const int row = p / (D / 8);
      const int c8 = (p - row * (D / 8)) * 8;
      const int qg = qt0 + row;
      bf16x8 qv = (bf16x8)(__bf16(0.f));
      bf16x8 dov = (bf16x8)(__bf16(0.f));
      if (qg < sq) {
        qv = *reinterpret_cast<const bf16x8*>(
            q + q_base + (long)qg * q_stride + c8);
        dov = *reinterpret_cast<const bf16x8*>(
            dout + q_base + (long)qg * q_stride + c8);
      }
      *reinterpret_cast<bf16x8*>(q_lds + row * QROW + c8) = qv;
      *reinterpret_cast<bf16x8*>(do_lds + row * QROW + c8) = dov;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        qt_lds[(c8 + i) * TROW + row] = qv[i];
        dot_lds[(c8 + i) * TROW + row] = dov[i];
      }
    }
    if (tid < KVB) {
      const int qg = qt0 + tid;
      lse_lds[tid] = qg < sq ? lse[((long)bi * hq + h) * sq + qg] : INFINITY;
      di_lds[tid] = qg < sq ? di[((long)bi * hq + h) * sq + qg] : 0.f;
    }
    __syncthreads();

    // S^T = K Q^T  and  dP^T = V dO^T
    f32x16 st_acc = (f32x16)(0.f), dpt_acc = (f32x16)(0.f);
#pragma unroll
    for (int ks = 0; ks < NK; ++ks) {
      bf16x8 qb = *reinterpret_cast<const bf16x8*>(
          q_lds + col * QROW + ks * 16 + 8 * hi);
      st_acc = mfma32_bf16(kf[ks], qb, st_acc);
      bf16x8 dob = *reinterpret_cast<const bf16x8*>(
          do_lds + col * QROW + ks * 16 + 8 * hi);
      dpt_acc = mfma32_bf16(vf[ks], dob, dpt_acc);
    }

    const int qg = qt0 + col;
    const float lse_c = lse_lds[col];
    const float di_c = di_lds[col];
    float pt[16];
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const int keyg = k0w + mfma32_d_row(lane, i);
      float p = 0.f;
      if (qg < sq && keyg < skv && !(causal && keyg > qg + off))
        p = __expf(st_acc[i] * scale - lse_c);
      pt[i] = p;
      my_p[mfma32_d_row(lane, i) * PROW + col] = (__bf16)p;
    }
    __syncthreads();  // none needed across waves for my_p; guards qt reuse

    // dV += P^T dO  (P^T via LDS A-frags, dO^T tile as B)
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          my_p + col * PROW + kh * 16 + 8 * hi);
#pragma unroll
      for (int dt = 0; dt < ND; ++dt) {
        bf16x8 dob = *reinterpret_cast<const bf16x8*>(
            dot_lds + (dt * 32 + col) * TROW + kh * 16 + 8 * hi);
        dv_acc[dt] = mfma32_bf16(pa, dob, dv_acc[dt]);
      }
    }

    // dS^T then dK += dS^T Q
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const float ds = pt[i] * (dpt_acc[i] - di_c) * scale;
      my_p[mfma32_d_row(lane, i) * PROW + col] = (__bf16)ds;
    }
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8 da = *reinterpret_cast<const bf16x8*>(
          my_p + col * PROW + kh * 16 + 8 * hi);
#pragma unroll
      for (int dt = 0; dt < ND; ++dt) {
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            qt_lds + (dt * 32 + col) * TROW + kh * 16 + 8 * hi);
        dk_acc[dt] = mfma32_bf16(da, qb, dk_acc[dt]);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int keyg = k0w + mfma32_d_row(lane, i);
    if (keyg < skv) {
      __bf16* dkr = dk_exp + dkv_base + (long)keyg * dkv_stride;
      __bf16* dvr = dv_exp + dkv_base + (long)keyg * dkv_stride;
#pragma unroll
      for (int dt = 0; dt < ND; ++dt) {
        dkr[dt * 32 + col] = (__bf16)dk_acc[dt][i];
        dvr[dt * 32 + col] = (__bf16)dv_acc[dt][i];
      }
    }
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
                               bool causal, hipStream_t st) {
  dim3 grid((sq + QB - 1) / QB, b * hq);
  hipLaunchKernelGGL((flash_fwd_kernel<D>), grid, dim3(256), 0, st, q, k, v,
                     o, lse, b, sq, skv, hq, hkv, scale, causal);
}

void flash_fwd_launch(const __bf16* q, const __bf16* k, const __bf16* v,
                      __bf16* o, float* lse, int b, int sq, int skv, int hq,
                      int hkv, int d, float scale, bool causal,
                      hipStream_t st) {
  if (d == 64)
    flash_fwd_launch_d<64>(q, k, v, o, lse, b, sq, skv, hq, hkv, scale,
                           causal, st);
  else
    flash_fwd_launch_d<128>(q, k, v, o, lse, b, sq, skv, hq, hkv, scale,
                            causal, st);
}

void attn_di_launch(const __bf16* dout, const __bf16* o, float* di, int b,
                    int sq, int hq, int d, hipStream_t st) {
  long rows = (long)b * sq * hq;
  int grid = galv_grid((rows + 3) / 4);
  if (d == 64)
    hipLaunchKernelGGL((attn_di_kernel<64>), dim3(grid), dim3(256), 0, st,
                       dout, o, di, b, sq, hq);
  else
    hipLaunchKernelGGL((attn_di_kernel<128>), dim3(grid), dim3(256), 0, st,
                       dout, o, di, b, sq, hq);
}

template <int D>
static void flash_bwd_launch_d(const __bf16* dout, const __bf16* q,
                               const __bf16* k, const __bf16* v,
                               const float* lse, const float* di, __bf16* dq,
                               __bf16* dk_exp, __bf16* dv_exp, int b, int sq,
                               int skv, int hq, int hkv, float scale,
                               bool causal, hipStream_t st) {
  dim3 gq((sq + QB - 1) / QB, b * hq);
  hipLaunchKernelGGL((flash_bwd_dq_kernel<D>), gq, dim3(256), 0, st, dout, q,
                     k, v, lse, di, dq, b, sq, skv, hq, hkv, scale, causal);
  dim3 gkv((skv + QB - 1) / QB, b * hq);
  hipLaunchKernelGGL((flash_bwd_dkv_kernel<D>), gkv, dim3(256), 0, st, dout,
                     q, k, v, lse, di, dk_exp, dv_exp, b, sq, skv, hq, hkv,
                     scale, causal);
}

void flash_bwd_launch(const __bf16* dout, const __bf16* q, const __bf16* k,
                      const __bf16* v, const float* lse, const float* di,
                      __bf16* dq, __bf16* dk_exp, __bf16* dv_exp, int b,
                      int sq, int skv, int hq, int hkv, int d, float scale,
                      bool causal, hipStream_t st) {
  if (d == 64)
    flash_bwd_launch_d<64>(dout, q, k, v, lse, di, dq, dk_exp, dv_exp, b, sq,
                           skv, hq, hkv, scale, causal, st);
  else
    flash_bwd_launch_d<128>(dout, q, k, v, lse, di, dq, dk_exp, dv_exp, b,
                            sq, skv, hq, hkv, scale, causal, st);
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
