// CDNA4 grouped GEMM for MoE experts (variable tokens per expert).
//
// Replaces the reference's external grouped_gemm CUDA dependency
// (galvatron/core/runtime/moe/grouped_gemm_util.py:1-22, used by
// GroupedMLP moe/mlp.py:26-120).
//
// Shapes: A [M_tot, K] bf16 row-major (expert-sorted tokens),
// W [E, K, N] bf16, C [M_tot, N]; per-expert row ranges from a host-built
// tile descriptor (expert id + row base per 128-row M tile).
//
// Structure (guide §5 anatomy): 128x128 output tile, BK=32, 4 waves each
// owning a 64x64 quadrant as 2x2 v_mfma_f32_32x32x16_bf16 accumulators;
// A staged row-major, W staged as an [n][k] image (column loads when the
// operand is W, direct rows when it is W^T) so the B fragment is one
// b128 read; double-buffered LDS, one barrier per K step.
//   TRANS_B=0: C = A @ W[e]        (expert forward / dX needs W^T -> 1)
//   TRANS_B=1: C = A @ W[e]^T      (W stored [N, K] for this call)
// dW = A^T dC is the separate grouped_gemm_dw kernel (contraction over
// the variable m dimension).
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;

struct TileDesc {
  int expert;
  int row0;    // first row of this M tile in A/C
  int rows;    // rows in this tile (<= 128)
};

template <bool TRANS_B>
__global__ __launch_bounds__(256, 2)
void grouped_gemm_kernel(const __bf16* __restrict__ A,
                         const __bf16* __restrict__ W,
                         __bf16* __restrict__ C,
                         const TileDesc* __restrict__ tiles,
                         int K, int N, long w_stride) {
  constexpr int BM = 128, BN = 128, BK = 32;
  constexpr int AROW = BK + 8;   // A image: [BM][BK+8]
  constexpr int WROW = BK + 8;   // W image: [BN][BK+8] (I[n][k])
  constexpr int BUFSZ = BM * AROW + BN * WROW;
  __shared__ __align__(16) __bf16 smem[2 * BUFSZ];

  const TileDesc td = tiles[blockIdx.x];
  const int n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;
  // wave quadrant: 2x2 of 64x64
  const int wr = (wid >> 1) * 64;   // row offset of the wave's quadrant
  const int wc = (wid & 1) * 64;    // col offset

  const __bf16* We = W + (long)td.expert * w_stride;
  const int ldw = TRANS_B ? K : N;  // leading dim of W memory rows

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  // staging registers (async split: load -> compute -> write)
  bf16x8 ast[2];    // 512 packs of A tile over 256 threads
  ushort8 wst[2];   // W image packs

  auto stage_load = [&](int k0) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;             // A: [BM][BK/8] packs
      const int row = idx / (BK / 8);
      const int c8 = (idx - row * (BK / 8)) * 8;
      const int rg = min(row, td.rows - 1);
      ast[p] = *reinterpret_cast<const bf16x8*>(
          A + (long)(td.row0 + rg) * K + k0 + c8);
      if (row >= td.rows) ast[p] = (bf16x8)(__bf16(0.f));
    }
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;             // W image: [BN][BK/8]
      const int n = idx / (BK / 8);
      const int kc = (idx - n * (BK / 8)) * 8;
      if (TRANS_B) {
        // Wsel[k][n] = W[n][k]: row n of W memory, contiguous k
        wst[p] = *reinterpret_cast<const ushort8*>(
            reinterpret_cast<const unsigned short*>(We) +
            (long)(n0 + n) * ldw + k0 + kc);
      } else {
        // Wsel[k][n] = W[k][n]: column loads (lane-coalesced over n)
        const unsigned short* wb =
            reinterpret_cast<const unsigned short*>(We) + n0 + n;
        ushort8 t;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          t[j] = wb[(long)(k0 + kc + j) * ldw];
        wst[p] = t;
      }
    }
  };
  auto stage_write = [&](int buf) {
    __bf16* a_lds = smem + buf * BUFSZ;
    __bf16* w_lds = a_lds + BM * AROW;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;
      const int row = idx / (BK / 8);
      const int c8 = (idx - row * (BK / 8)) * 8;
      *reinterpret_cast<bf16x8*>(a_lds + row * AROW + c8) = ast[p];
    }
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;
      const int n = idx / (BK / 8);
      const int kc = (idx - n * (BK / 8)) * 8;
      *reinterpret_cast<ushort8*>(w_lds + n * WROW + kc) = wst[p];
    }
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  int cur = 0;

  for (int k0 = 0; k0 < K; k0 += BK) {
    const bool have_next = k0 + BK < K;
    const __bf16* a_lds = smem + cur * BUFSZ;
    const __bf16* w_lds = a_lds + BM * AROW;
    if (have_next) stage_load(k0 + BK);

#pragma unroll
    for (int ks = 0; ks < BK / 16; ++ks) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        bf16x8 af = *reinterpret_cast<const bf16x8*>(
            a_lds + (wr + i * 32 + col) * AROW + ks * 16 + 8 * hi);
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          bf16x8 wf = *reinterpret_cast<const bf16x8*>(
              w_lds + (wc + j * 32 + col) * WROW + ks * 16 + 8 * hi);
          // A-frag must pair with the wave's own row; re-read per (i, j)
          // is avoided by hoisting af above. D = A-rows x W-cols:
          acc[i][j] = mfma32_bf16(af, wf, acc[i][j]);
        }
      }
    }
    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: D-layout rows=m(crow), cols=n(lane)  — wait: swapped!
  // mfma(A_frag, B_frag): A rows (m) spread over regs, B col (n) per lane.
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int m = wr + i * 32 + mfma32_d_row(lane, r);
        const int n = wc + j * 32 + col;
        if (m < td.rows)
          C[(long)(td.row0 + m) * N + n0 + n] = (__bf16)acc[i][j][r];
      }
}

// dW[e] = A_e^T dC_e  — grid (m-tile-independent): (expert, K/128, N/128),
// contraction over the expert's variable m rows in 32-row chunks.
// Both operand images are [out-dim][m] built by column loads.
__global__ __launch_bounds__(256, 2)
void grouped_gemm_dw_kernel(const __bf16* __restrict__ A,
                            const __bf16* __restrict__ dC,
                            float* __restrict__ dW,   // [E, K, N] fp32 accum
                            const int* __restrict__ row_off,  // [E+1]
                            int K, int N) {
  constexpr int BKD = 128;  // output K rows per block
  constexpr int BN = 128;
  constexpr int BM = 32;    // contraction chunk
  constexpr int TROW = BM + 8;
  constexpr int BUFSZ = BKD * TROW + BN * TROW;
  __shared__ __align__(16) __bf16 smem[2 * BUFSZ];

  const int e = blockIdx.x;
  const int kk0 = blockIdx.y * BKD;
  const int n0 = blockIdx.z * BN;
  const int m_lo = row_off[e], m_hi = row_off[e + 1];
  const int m_cnt = m_hi - m_lo;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;
  const int wr = (wid >> 1) * 64;
  const int wc = (wid & 1) * 64;

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.f);

  ushort8 at_st[2], ct_st[2];
  auto stage_load = [&](int m0) {
    const bool full = (m0 + BM <= m_cnt);
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;           // [BKD][BM/8] packs
      const int kr = idx / (BM / 8);
      const int mc = (idx - kr * (BM / 8)) * 8;
      const unsigned short* ab =
          reinterpret_cast<const unsigned short*>(A) + kk0 + kr;
      ushort8 ta, tc;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int m = full ? m0 + mc + j : min(m0 + mc + j, m_cnt - 1);
        ta[j] = ab[(long)(m_lo + m) * K];
        unsigned short v = reinterpret_cast<const unsigned short*>(dC)
            [(long)(m_lo + m) * N + n0 + kr];
        tc[j] = v;
        if (!full && m0 + mc + j >= m_cnt) { ta[j] = 0; tc[j] = 0; }
      }
      at_st[p] = ta;
      ct_st[p] = tc;
    }
  };
  auto stage_write = [&](int buf) {
    __bf16* at_lds = smem + buf * BUFSZ;
    __bf16* ct_lds = at_lds + BKD * TROW;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int idx = tid + p * 256;
      const int kr = idx / (BM / 8);
      const int mc = (idx - kr * (BM / 8)) * 8;
      *reinterpret_cast<ushort8*>(at_lds + kr * TROW + mc) = at_st[p];
      *reinterpret_cast<ushort8*>(ct_lds + kr * TROW + mc) = ct_st[p];
    }
  };

  if (m_cnt > 0) {
  stage_load(0);
  stage_write(0);
  __syncthreads();
  int cur = 0;

  for (int m0 = 0; m0 < m_cnt; m0 += BM) {
    const bool have_next = m0 + BM < m_cnt;
    const __bf16* at_lds = smem + cur * BUFSZ;
    const __bf16* ct_lds = at_lds + BKD * TROW;
    if (have_next) stage_load(m0 + BM);

#pragma unroll
    for (int ks = 0; ks < BM / 16; ++ks) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        bf16x8 af = *reinterpret_cast<const bf16x8*>(
            at_lds + (wr + i * 32 + col) * TROW + ks * 16 + 8 * hi);
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          bf16x8 cf = *reinterpret_cast<const bf16x8*>(
              ct_lds + (wc + j * 32 + col) * TROW + ks * 16 + 8 * hi);
          acc[i][j] = mfma32_bf16(af, cf, acc[i][j]);
        }
      }
    }
    if (have_next) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }
  }  // m_cnt > 0

#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kk = wr + i * 32 + mfma32_d_row(lane, r);
        const int n = wc + j * 32 + col;
        dW[((long)e * K + kk0 + kk) * N + n0 + n] = acc[i][j][r];
      }
}

}  // namespace

void grouped_gemm_launch(const __bf16* A, const __bf16* W, __bf16* C,
                         const void* tiles, int n_tiles, int K, int N,
                         long w_stride, bool trans_b, hipStream_t st) {
  dim3 grid(n_tiles, N / 128);
  if (trans_b)
    hipLaunchKernelGGL((grouped_gemm_kernel<true>), grid, dim3(256), 0, st,
                       A, W, C, (const TileDesc*)tiles, K, N, w_stride);
  else
    hipLaunchKernelGGL((grouped_gemm_kernel<false>), grid, dim3(256), 0, st,
                       A, W, C, (const TileDesc*)tiles, K, N, w_stride);
}

void grouped_gemm_dw_launch(const __bf16* A, const __bf16* dC, float* dW,
                            const int* row_off, int E, int K, int N,
                            hipStream_t st) {
  dim3 grid(E, K / 128, N / 128);
  hipLaunchKernelGGL(grouped_gemm_dw_kernel, grid, dim3(256), 0, st, A, dC,
                     dW, row_off, K, N);
}
