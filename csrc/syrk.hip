// Fused covariance (SYRK) kernels for K-FAC factor accumulation.
//
// Computes out = beta*out + coeff * (A^T A) for three virtual matrices A
// without ever materializing them in HBM:
//   - LinearAcc:    A = [activations, ones] (bias column fused)
//   - ConvPatchAcc: A = im2col(x) patches [+ ones]  (K2+K3 of SURVEY §2.4)
//   - ConvGradAcc:  A = NCHW -> (N*OH*OW, C) transpose view
//
// MI355X design notes:
//   * mfma_f32_16x16x4_f32: exact f32 numerics (matches the fp32 torch
//     reference bit-for-bit in summation class), 157 TF chip peak.
//   * C = A^T A is symmetric: only tiles ti<=tj are computed; off-diagonal
//     tiles are written twice (mirrored), making the output exactly
//     symmetric by construction (the reference symmetrizes explicitly,
//     kfac/layers/utils.py:55-57).
//   * The reduction (batch) dimension M is split across blockIdx.z so even
//     a single-tile factor (64x64) launches enough workgroups to fill 256
//     CUs; partial tiles accumulate with fp32 atomics (bytes are trivial
//     against the MFMA work).
//   * bf16/f16 inputs are converted in the LDS staging pass and accumulated
//     in fp32 (better numerics than the reference's bf16 matmul).

#include "common.h"

#include <type_traits>

namespace kfac {

constexpr int BT = 64;   // C tile edge
constexpr int BK = 32;   // m-slice staged per iteration (f32 path)
constexpr int LDS_STRIDE = BT + 1;  // break bank alignment for b32 reads

// bf16 path: K=32-deep MFMA, 64-deep m-slices, [col][m] LDS layout so each
// lane's 8-element fragment is one 16-byte ds_read_b128.
constexpr int BKB = 64;
constexpr int BSTR = BKB + 8;  // row stride 144 B: 16-lane groups hit 16
                               // distinct banks ((144/4)*j mod 64, gcd 4)
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

// k-index swizzle for the bf16 [col][m] LDS tiles: rows 8 apart share
// write banks at the 144 B stride (measured 7.2e8 SQ_LDS_BANK_CONFLICT
// on the LinearAcc SYRK); XOR-ing k by a per-row constant in 16-bf16
// (32 B) units keeps every 8-element fragment contiguous and 16 B
// aligned while spreading the colfast scalar writes over banks.
__device__ __forceinline__ int kswz(int row, int k) {
  // With the transposed staging images every LDS write is a contiguous
  // 16 B store whose 8-lane write group spans <= 4 consecutive rows
  // (conflict-free at the 144 B stride by itself); on the READ side a
  // per-16-row-block constant measured 43% fewer conflict cycles than
  // the earlier (row>>3)+(row>>5) form (9.4e8 -> 5.4e8 on the cov
  // sweep, PMC). The residual comes from ds_read_b128's scrambled
  // 16-lane service groups (MI355X_MICROARCH §LDS) mixing rows from two
  // kfrag quarters; eliminating it did not change kernel time, so the
  // simple form stays.
  return k ^ (((row >> 4) & 3) << 4);
}

// ---------------------------------------------------------------- accessors

template <typename T>
struct LinearAcc {
  const T* a;
  long lda;
  int M;       // rows
  int K;       // real columns (bias column at index K if Ncols == K+1)
  int Ncols;   // K + (bias ? 1 : 0)
  static constexpr bool kLaneAlongCols = true;

  __device__ __forceinline__ float load(int m, int i) const {
    if (m >= M || i >= Ncols) return 0.0f;
    if (i < K) return to_f32(a[(long)m * lda + i]);
    return 1.0f;  // fused bias-ones column
  }

  // 16 consecutive-m values at a fixed column (bf16-path staging).
  __device__ __forceinline__ void load16(int m, int i, float* out) const {
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = load(m + e, i);
  }

  __device__ __forceinline__ void load16_bf16(int m, int i, __bf16* out) const {
    float tmp[16];
    load16(m, i, tmp);
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = (__bf16)tmp[e];
  }

  // 16 consecutive COLUMNS of one row: contiguous 32 B of HBM.
  __device__ __forceinline__ void load16cols_bf16(int m, int i0, __bf16* out) const {
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      if (m < M && i0 + 15 < K) {
        __builtin_memcpy(out, a + (long)m * lda + i0, 16 * sizeof(__bf16));
        return;
      }
    }
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = (__bf16)load(m, i0 + e);
  }
};

// Transposed-image accessor: the matrix is materialized as [K][Mpad]
// (m fastest-varying), so 16 consecutive-m values of one factor column
// are 32 contiguous bytes of HBM and the m-fast staging path (contiguous
// 16 B LDS writes, kLaneAlongCols=false) applies. This replaced the
// column-fast LinearAcc staging for bf16: that path issued 16 scalar LDS
// stores per 16 values (8192 per 128x64 tile stage, ~8x the MFMA issue
// cycles of the slice). Rows m in [M, Mpad) are zero-filled by the
// producer kernels so wide loads may cross M; the bias-ones column is
// synthesized with the exact m < M bound (it accumulates the count).
template <typename T>
struct TransAcc {
  const T* a;  // [K][Mpad]
  int Mpad;
  int M;      // real rows (loop bound; bias-ones extent)
  int K;      // materialized columns (= rows of the transposed image)
  int Ncols;  // K + (bias ? 1 : 0)
  static constexpr bool kLaneAlongCols = false;

  __device__ __forceinline__ float load(int m, int i) const {
    if (m >= M || i >= Ncols) return 0.0f;
    if (i < K) return to_f32(a[(long)i * Mpad + m]);
    return 1.0f;
  }

  __device__ __forceinline__ void load16(int m, int i, float* out) const {
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = load(m + e, i);
  }

  __device__ __forceinline__ void load16_bf16(int m, int i, __bf16* out) const {
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      if (i < K && m + 15 < Mpad) {
        __builtin_memcpy(out, a + (long)i * Mpad + m, 16 * sizeof(__bf16));
        return;
      }
    }
    if (i >= K && i < Ncols) {
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        out[e] = (__bf16)((m + e < M) ? 1.0f : 0.0f);
      }
      return;
    }
    float tmp[16];
    load16(m, i, tmp);
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = (__bf16)tmp[e];
  }
};

template <typename T>
struct ConvPatchAcc {
  const T* x;
  int C, H, W, OH, OW;
  int kh, kw, sh, sw, ph, pw;
  int M;     // Nb*OH*OW
  int K;     // C*kh*kw
  int Ncols;
  static constexpr bool kLaneAlongCols = false;  // lanes along m (w-contig)

  __device__ __forceinline__ float load(int m, int i) const {
    if (m >= M || i >= Ncols) return 0.0f;
    if (i >= K) return 1.0f;
    int ow = m % OW;
    int t = m / OW;
    int oh = t % OH;
    int n = t / OH;
    int s = i % kw;
    int t2 = i / kw;
    int r = t2 % kh;
    int c = t2 / kh;
    int h = oh * sh - ph + r;
    int w = ow * sw - pw + s;
    if (h < 0 || h >= H || w < 0 || w >= W) return 0.0f;
    return to_f32(x[((long)(n * C + c) * H + h) * W + w]);
  }

  // 16 consecutive-m (= consecutive ow) values at a fixed patch column:
  // one div/mod decomposition, then incremental ow/h walk. Consecutive m
  // is stride-sw in w, so loads are HBM-contiguous for stride-1 convs.
  __device__ __forceinline__ void load16(int m, int i, float* out) const {
    if (i >= K || m >= M) {
      // bias-ones column (i == K < Ncols) or out-of-range padding
      float fill = (i >= K && i < Ncols && m < M) ? 1.0f : 0.0f;
#pragma unroll
      for (int e = 0; e < 16; ++e) out[e] = (m + e < M) ? fill : 0.0f;
      return;
    }
    int s = i % kw;
    int t2 = i / kw;
    int r = t2 % kh;
    int c = t2 / kh;
    int ow = m % OW;
    int t = m / OW;
    int oh = t % OH;
    int n = t / OH;
    int h = oh * sh - ph + r;
    int w = ow * sw - pw + s;
    const T* base = x + ((long)(n * C + c) * H) * W;
#pragma unroll
    for (int e = 0; e < 16; ++e) {
      bool ok = (m + e < M) && h >= 0 && h < H && w >= 0 && w < W;
      out[e] = ok ? to_f32(base[(long)h * W + w]) : 0.0f;
      // advance to the next output position
      ++ow;
      w += sw;
      if (ow == OW) {
        ow = 0;
        w = -pw + s;
        ++oh;
        h += sh;
        if (oh == OH) {
          oh = 0;
          h = -ph + r;
          ++n;
          base += (long)C * H * W;
        }
      }
    }
  }

  // bf16 staging fast path: for unit-stride interior runs the 16 values
  // are 32 contiguous bytes of HBM -> two (possibly unaligned) 16-byte
  // loads instead of 16 scalar gathers.
  __device__ __forceinline__ void load16_bf16(int m, int i, __bf16* out) const {
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      if (i < K && m + 15 < M) {
        int s = i % kw;
        int t2 = i / kw;
        int r = t2 % kh;
        int c = t2 / kh;
        int ow = m % OW;
        int t = m / OW;
        int oh = t % OH;
        int h = oh * sh - ph + r;
        int w = ow * sw - pw + s;
        if (sw == 1 && ow + 15 < OW && h >= 0 && h < H && w >= 0 &&
            w + 15 < W) {
          int n = t / OH;
          const T* p = x + ((long)(n * C + c) * H + h) * W + w;
          __builtin_memcpy(out, p, 16 * sizeof(__bf16));
          return;
        }
        if (sw == 2 && ow + 15 < OW && h >= 0 && h < H && w >= 0 &&
            w + 31 < W) {
          // stride-2 interior: 16 values span 31 contiguous elements ->
          // two 32-byte wide loads, keep every other element.
          int n = t / OH;
          const T* p = x + ((long)(n * C + c) * H + h) * W + w;
          __bf16 tmp[32];
          __builtin_memcpy(tmp, p, 32 * sizeof(__bf16));
#pragma unroll
          for (int e = 0; e < 16; ++e) out[e] = tmp[2 * e];
          return;
        }
      }
    }
    float tmp[16];
    load16(m, i, tmp);
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = (__bf16)tmp[e];
  }
};

template <typename T>
struct ConvGradAcc {
  const T* g;
  int C, OH, OW;
  int M;  // Nb*OH*OW
  int Ncols;
  static constexpr bool kLaneAlongCols = false;

  __device__ __forceinline__ float load(int m, int j) const {
    if (m >= M || j >= Ncols) return 0.0f;
    int ow = m % OW;
    int t = m / OW;
    int oh = t % OH;
    int n = t / OH;
    return to_f32(g[((long)(n * C + j) * OH + oh) * OW + ow]);
  }

  // 16 consecutive-m values of one channel: contiguous in (oh, ow) except
  // at image boundaries -> one decomposition + pointer walk.
  __device__ __forceinline__ void load16(int m, int j, float* out) const {
    if (m >= M || j >= Ncols) {
#pragma unroll
      for (int e = 0; e < 16; ++e) out[e] = 0.0f;
      return;
    }
    int sp = m % (OH * OW);  // position within the image plane
    int n = m / (OH * OW);
    const T* p = g + ((long)(n * C + j) * OH * OW) + sp;
#pragma unroll
    for (int e = 0; e < 16; ++e) {
      out[e] = (m + e < M) ? to_f32(*p) : 0.0f;
      ++sp;
      ++p;
      if (sp == OH * OW) {
        sp = 0;
        p += (long)(C - 1) * OH * OW;
      }
    }
  }

  __device__ __forceinline__ void load16_bf16(int m, int j, __bf16* out) const {
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      if (j < Ncols && m + 15 < M) {
        int sp = m % (OH * OW);
        if (sp + 15 < OH * OW) {
          int n = m / (OH * OW);
          const T* p = g + ((long)(n * C + j) * OH * OW) + sp;
          __builtin_memcpy(out, p, 16 * sizeof(__bf16));
          return;
        }
      }
    }
    float tmp[16];
    load16(m, j, tmp);
#pragma unroll
    for (int e = 0; e < 16; ++e) out[e] = (__bf16)tmp[e];
  }
};

// ---------------------------------------------------------------- kernel

template <typename Acc>
__device__ __forceinline__ void stage_tile(
    const Acc& acc,
    float (*lds)[LDS_STRIDE],
    int m0,
    int col0,
    int tid) {
  // 32 x 64 elements, 256 threads -> 8 each.
#pragma unroll
  for (int e = 0; e < (BK * BT) / 256; ++e) {
    int idx = tid + e * 256;
    int k, i;
    if (Acc::kLaneAlongCols) {
      k = idx / BT;
      i = idx % BT;
    } else {
      k = idx % BK;
      i = idx / BK;
    }
    lds[k][i] = acc.load(m0 + k, col0 + i);
  }
}

template <typename AccL, typename AccR>
__global__ __launch_bounds__(256) void syrk_kernel(
    float* __restrict__ out,
    int N,            // factor edge (= Ncols)
    AccL accl,
    AccR accr,
    int m_per_split,
    float coeff,
    bool same_tile_ok  // accl and accr view the same matrix
) {
  const int ti = blockIdx.x;
  const int tj = blockIdx.y;
  if (tj < ti) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // wave row (0..1) -> 32-row quadrant
  const int wc = wave & 1;   // wave col

  const int i0 = ti * BT;
  const int j0 = tj * BT;
  const int m_begin = blockIdx.z * m_per_split;
  const int m_end = min(accl.M, m_begin + m_per_split);
  if (m_begin >= m_end) return;

  __shared__ float lds_l[BK][LDS_STRIDE];
  __shared__ float lds_r[BK][LDS_STRIDE];
  const bool diag = same_tile_ok && (ti == tj);

  f32x4 acc[2][2] = {};

  for (int m0 = m_begin; m0 < m_end; m0 += BK) {
    stage_tile(accl, lds_l, m0, i0, tid);
    if (!diag) {
      stage_tile(accr, lds_r, m0, j0, tid);
    }
    __syncthreads();
    auto rbuf = diag ? lds_l : lds_r;
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const int krow = kk + (lane >> 4);
      float a0 = lds_l[krow][wr * 32 + (lane & 15)];
      float a1 = lds_l[krow][wr * 32 + 16 + (lane & 15)];
      float b0 = rbuf[krow][wc * 32 + (lane & 15)];
      float b1 = rbuf[krow][wc * 32 + 16 + (lane & 15)];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D fragment layout (16x16): row = (lane>>4)*4 + r, col = lane&15.
  const bool mirror = (ti != tj);
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i0 + wr * 32 + fi * 16 + (lane >> 4) * 4 + r;
        int col = j0 + wc * 32 + fj * 16 + (lane & 15);
        if (row < N && col < N) {
          float v = coeff * acc[fi][fj][r];
          atomicAdd(&out[(long)row * N + col], v);
          if (mirror) {
            atomicAdd(&out[(long)col * N + row], v);
          }
        }
      }
    }
  }
}

// bf16 variant: inputs staged to LDS as bf16 in [col][m] layout so each
// lane's 8-element MFMA fragment is one 16-byte ds_read_b128; compute on
// mfma_f32_16x16x32_bf16 (fp32 accumulate), ~16x the f32-MFMA rate.
// Column-fast staging (LinearAcc): each thread reads 16 contiguous
// columns of one m-row from HBM and scatters them down one LDS column.
template <typename Acc>
__device__ __forceinline__ void stage_tile_bf16_colfast(
    const Acc& acc,
    __bf16 (*lds)[BSTR],
    int m0,
    int col0,
    int tid) {
  const int k = tid >> 2;
  const int i0 = (tid & 3) * 16;
  __bf16 vals[16];
  acc.load16cols_bf16(m0 + k, col0 + i0, vals);
#pragma unroll
  for (int e = 0; e < 16; ++e) {
    lds[i0 + e][kswz(i0 + e, k)] = vals[e];
  }
}

template <typename Acc>
__device__ __forceinline__ void stage_tile_bf16(
    const Acc& acc,
    __bf16 (*lds)[BSTR],
    int m0,
    int col0,
    int tid) {
  // 64 cols x 64 m-values, 256 threads: each thread stages one 16-deep
  // m-run of one column (4 threads per column). The 16 bf16 land as two
  // 16-byte LDS writes (row stride 144 B keeps them aligned).
  const int i = tid >> 2;
  const int k0 = (tid & 3) * 16;
  __bf16 vals[16];
  acc.load16_bf16(m0 + k0, col0 + i, vals);
  *(bf16x8*)&lds[i][kswz(i, k0)] = *(const bf16x8*)&vals[0];
  *(bf16x8*)&lds[i][kswz(i, k0 + 8)] = *(const bf16x8*)&vals[8];
}

template <typename AccL, typename AccR>
__global__ __launch_bounds__(256) void syrk_kernel_bf16(
    float* __restrict__ out,
    int N,
    AccL accl,
    AccR accr,
    int m_per_split,
    float coeff,
    bool same_tile_ok) {
  const int ti = blockIdx.x;
  const int tj = blockIdx.y;
  if (tj < ti) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int i0 = ti * BT;
  const int j0 = tj * BT;
  const int m_begin = blockIdx.z * m_per_split;
  const int m_end = min(accl.M, m_begin + m_per_split);
  if (m_begin >= m_end) return;

  __shared__ __bf16 lds_l[BT][BSTR];
  __shared__ __bf16 lds_r[BT][BSTR];
  const bool diag = same_tile_ok && (ti == tj);

  f32x4 acc[2][2] = {};

  for (int m0 = m_begin; m0 < m_end; m0 += BKB) {
    if constexpr (AccL::kLaneAlongCols) {
      stage_tile_bf16_colfast(accl, lds_l, m0, i0, tid);
      if (!diag) stage_tile_bf16_colfast(accr, lds_r, m0, j0, tid);
    } else {
      stage_tile_bf16(accl, lds_l, m0, i0, tid);
      if (!diag) stage_tile_bf16(accr, lds_r, m0, j0, tid);
    }
    __syncthreads();
    auto rbuf = diag ? lds_l : lds_r;
#pragma unroll
    for (int kt = 0; kt < BKB; kt += 32) {
      const int kfrag = kt + (lane >> 4) * 8;
      const int r0 = wr * 32 + (lane & 15);
      const int r1 = r0 + 16;
      const int c0 = wc * 32 + (lane & 15);
      const int c1 = c0 + 16;
      bf16x8 a0 = *(const bf16x8*)&lds_l[r0][kswz(r0, kfrag)];
      bf16x8 a1 = *(const bf16x8*)&lds_l[r1][kswz(r1, kfrag)];
      bf16x8 b0 = *(const bf16x8*)&rbuf[c0][kswz(c0, kfrag)];
      bf16x8 b1 = *(const bf16x8*)&rbuf[c1][kswz(c1, kfrag)];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  const bool mirror = (ti != tj);
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i0 + wr * 32 + fi * 16 + (lane >> 4) * 4 + r;
        int col = j0 + wc * 32 + fj * 16 + (lane & 15);
        if (row < N && col < N) {
          float v = coeff * acc[fi][fj][r];
          atomicAdd(&out[(long)row * N + col], v);
          if (mirror) {
            atomicAdd(&out[(long)col * N + row], v);
          }
        }
      }
    }
  }
}


// 128x128-tile bf16 SYRK for large factors (N > 128): 4 waves each own a
// 64x64 quadrant as 4x4 fragments (64 fp32 accumulators/lane) -> 32
// MFMAs per 64-deep m-slice per wave against 4 load16 staging calls.
constexpr int BTB = 128;

template <typename Acc>
__device__ __forceinline__ void stage_tile_bf16_big(
    const Acc& acc,
    __bf16 (*lds)[BSTR],
    int m0,
    int col0,
    int tid) {
  // 128 cols x 64 m, 256 threads: each thread stages two 16-deep m-runs
  // of one column.
  const int i = tid >> 1;
  const int k0 = (tid & 1) * 32;
  __bf16 vals[16];
  acc.load16_bf16(m0 + k0, col0 + i, vals);
  *(bf16x8*)&lds[i][kswz(i, k0)] = *(const bf16x8*)&vals[0];
  *(bf16x8*)&lds[i][kswz(i, k0 + 8)] = *(const bf16x8*)&vals[8];
  acc.load16_bf16(m0 + k0 + 16, col0 + i, vals);
  *(bf16x8*)&lds[i][kswz(i, k0 + 16)] = *(const bf16x8*)&vals[0];
  *(bf16x8*)&lds[i][kswz(i, k0 + 24)] = *(const bf16x8*)&vals[8];
}

template <typename Acc>
__device__ __forceinline__ void stage_tile_bf16_big_colfast(
    const Acc& acc,
    __bf16 (*lds)[BSTR],
    int m0,
    int col0,
    int tid) {
  const int k = tid >> 2;
  const int i0 = (tid & 3) * 32;
  __bf16 vals[16];
  acc.load16cols_bf16(m0 + k, col0 + i0, vals);
#pragma unroll
  for (int e = 0; e < 16; ++e) lds[i0 + e][kswz(i0 + e, k)] = vals[e];
  acc.load16cols_bf16(m0 + k, col0 + i0 + 16, vals);
#pragma unroll
  for (int e = 0; e < 16; ++e) {
    lds[i0 + 16 + e][kswz(i0 + 16 + e, k)] = vals[e];
  }
}

template <typename AccL, typename AccR>
__global__ __launch_bounds__(256) void syrk_kernel_bf16_big(
    float* __restrict__ out,
    int N,
    AccL accl,
    AccR accr,
    int m_per_split,
    float coeff,
    bool same_tile_ok) {
  const int ti = blockIdx.x;
  const int tj = blockIdx.y;
  if (tj < ti) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int i0 = ti * BTB;
  const int j0 = tj * BTB;
  const int m_begin = blockIdx.z * m_per_split;
  const int m_end = min(accl.M, m_begin + m_per_split);
  if (m_begin >= m_end) return;

  // Double-buffered slabs: slice t+1 stages into slab p^1 while slice
  // t's MFMAs read slab p — one barrier per m-slice.  The kernel is
  // HBM-streaming-bound (~55% of peak single-buffered); overlapping
  // the staging with the MFMAs recovers most of the latency.
  __shared__ __bf16 lds_l[2][BTB][BSTR];
  __shared__ __bf16 lds_r[2][BTB][BSTR];
  const bool diag = same_tile_ok && (ti == tj);

  f32x4 acc[4][4] = {};

  auto stage_into = [&](int slab, int m0) {
    if constexpr (AccL::kLaneAlongCols) {
      stage_tile_bf16_big_colfast(accl, lds_l[slab], m0, i0, tid);
      if (!diag) stage_tile_bf16_big_colfast(accr, lds_r[slab], m0, j0, tid);
    } else {
      stage_tile_bf16_big(accl, lds_l[slab], m0, i0, tid);
      if (!diag) stage_tile_bf16_big(accr, lds_r[slab], m0, j0, tid);
    }
  };

  stage_into(0, m_begin);
  __syncthreads();
  int p = 0;
  for (int m0 = m_begin; m0 < m_end; m0 += BKB) {
    if (m0 + BKB < m_end) {
      stage_into(p ^ 1, m0 + BKB);
    }
    auto rbuf = diag ? lds_l[p] : lds_r[p];
#pragma unroll
    for (int kt = 0; kt < BKB; kt += 32) {
      const int kfrag = kt + (lane >> 4) * 8;
      bf16x8 av[4];
      bf16x8 bv[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int ra = wr * 64 + f * 16 + (lane & 15);
        const int rb = wc * 64 + f * 16 + (lane & 15);
        av[f] = *(const bf16x8*)&lds_l[p][ra][kswz(ra, kfrag)];
        bv[f] = *(const bf16x8*)&rbuf[rb][kswz(rb, kfrag)];
      }
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av[fi], bv[fj], acc[fi][fj], 0, 0, 0);
        }
      }
    }
    __syncthreads();
    p ^= 1;
  }

  const bool mirror = (ti != tj);
#pragma unroll
  for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i0 + wr * 64 + fi * 16 + (lane >> 4) * 4 + r;
        int col = j0 + wc * 64 + fj * 16 + (lane & 15);
        if (row < N && col < N) {
          float v = coeff * acc[fi][fj][r];
          atomicAdd(&out[(long)row * N + col], v);
          if (mirror) {
            atomicAdd(&out[(long)col * N + row], v);
          }
        }
      }
    }
  }
}

__global__ void scale_kernel(float* out, long n, float beta) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    out[i] = (beta == 0.0f) ? 0.0f : out[i] * beta;
  }
}

// -------------------------------------------------------------- im2col
//
// Why materialize at all? PMC on the fused-gather SYRK showed 523 VALU
// per MFMA: the per-element patch address math re-runs for every
// (tile-row, tile-col) pair touching the element (~2x tiles per dim), so
// gather VALU — not HBM — dominated. Materializing runs the gather ONCE
// (coalesced 16-byte writes along K) and the SYRK then stages from the
// flat matrix with plain wide loads; the extra HBM round trip is cheap
// on 8 TB/s HBM3E relative to the n^2 MFMA work.
template <typename T>
__global__ __launch_bounds__(256) void im2col_kernel(
    T* __restrict__ dst,  // [M][K_pad]
    ConvPatchAcc<T> acc,
    int K_pad) {
  const int k8 = K_pad / 8;
  const long total = (long)acc.M * k8;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int m = (int)(idx / k8);
    const int k0 = (int)(idx % k8) * 8;
    // one m decomposition per 8 elements
    const int ow = m % acc.OW;
    int t = m / acc.OW;
    const int oh = t % acc.OH;
    const int n = t / acc.OH;
    T vals[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int k = k0 + e;
      T v = T(0.0f);
      if (k < acc.K) {
        const int s = k % acc.kw;
        const int t2 = k / acc.kw;
        const int r = t2 % acc.kh;
        const int c = t2 / acc.kh;
        const int h = oh * acc.sh - acc.ph + r;
        const int w = ow * acc.sw - acc.pw + s;
        if (h >= 0 && h < acc.H && w >= 0 && w < acc.W) {
          v = acc.x[((long)(n * acc.C + c) * acc.H + h) * acc.W + w];
        }
      }
      vals[e] = v;
    }
    __builtin_memcpy(&dst[(long)m * K_pad + k0], vals, 8 * sizeof(T));
  }
}

template <typename T>
hipError_t im2col_t(
    hipStream_t stream,
    const T* x,
    T* dst,
    int Nb,
    int C,
    int H,
    int W,
    int kh,
    int kw,
    int sh,
    int sw,
    int ph,
    int pw,
    int K_pad) {
  int OH = (H + 2 * ph - kh) / sh + 1;
  int OW = (W + 2 * pw - kw) / sw + 1;
  int K = C * kh * kw;
  ConvPatchAcc<T> acc{x,  C,  H,  W,  OH, OW, kh,
                      kw, sh, sw, ph, pw, Nb * OH * OW,
                      K,  K};
  long total = (long)acc.M * (K_pad / 8);
  int blocks = (int)min((total + 255) / 256, (long)2048);
  im2col_kernel<T><<<blocks, 256, 0, stream>>>(dst, acc, K_pad);
  return hipGetLastError();
}

template hipError_t im2col_t<float>(hipStream_t, const float*, float*, int, int, int, int, int, int, int, int, int, int, int);
template hipError_t im2col_t<__hip_bfloat16>(hipStream_t, const __hip_bfloat16*, __hip_bfloat16*, int, int, int, int, int, int, int, int, int, int, int);
template hipError_t im2col_t<__half>(hipStream_t, const __half*, __half*, int, int, int, int, int, int, int, int, int, int, int);

// Transposed im2col: dst is [K][Mpad] (m fastest-varying), the image the
// TransAcc SYRK staging wants. Each thread gathers 16 consecutive output
// positions of one patch column (one incremental ow/h walk) and writes
// one contiguous 32 B run; m in [M, Mpad) is zero-filled.
template <typename T>
__global__ __launch_bounds__(256) void im2col_tr_kernel(
    T* __restrict__ dst,  // [K][Mpad]
    ConvPatchAcc<T> acc,
    int Mpad) {
  const int m16 = Mpad / 16;
  const long total = (long)acc.K * m16;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int k = (int)(idx / m16);
    const int m0 = (int)(idx % m16) * 16;
    T vals[16];
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      acc.load16_bf16(m0, k, (__bf16*)vals);
    } else {
      float tmp[16];
      acc.load16(m0, k, tmp);
#pragma unroll
      for (int e = 0; e < 16; ++e) vals[e] = (T)tmp[e];
    }
    __builtin_memcpy(&dst[(long)k * Mpad + m0], vals, 16 * sizeof(T));
  }
}

template <typename T>
hipError_t im2col_tr_t(
    hipStream_t stream,
    const T* x,
    T* dst,
    int Nb,
    int C,
    int H,
    int W,
    int kh,
    int kw,
    int sh,
    int sw,
    int ph,
    int pw,
    int Mpad) {
  int OH = (H + 2 * ph - kh) / sh + 1;
  int OW = (W + 2 * pw - kw) / sw + 1;
  int K = C * kh * kw;
  ConvPatchAcc<T> acc{x,  C,  H,  W,  OH, OW, kh,
                      kw, sh, sw, ph, pw, Nb * OH * OW,
                      K,  K};
  long total = (long)K * (Mpad / 16);
  int blocks = (int)min((total + 255) / 256, (long)2048);
  im2col_tr_kernel<T><<<blocks, 256, 0, stream>>>(dst, acc, Mpad);
  return hipGetLastError();
}

template hipError_t im2col_tr_t<__hip_bfloat16>(hipStream_t, const __hip_bfloat16*, __hip_bfloat16*, int, int, int, int, int, int, int, int, int, int, int);

// [M][lda] row-major -> [K][Mpad] bf16, LDS-tiled 64x64 transpose (both
// global phases fully coalesced; runs once per factor so its cost is the
// one extra HBM round trip argued for materialized im2col above).
template <typename T>
__global__ __launch_bounds__(256) void transpose_to_bf16_kernel(
    const T* __restrict__ src,
    long lda,
    int M,
    int K,
    __bf16* __restrict__ dst,
    int Mpad) {
  __shared__ __bf16 tile[64][72];
  const int nmt = Mpad / 64;
  const int nkt = ceil_div(K, 64);
  const int tid = threadIdx.x;
  for (int t = blockIdx.x; t < nmt * nkt; t += gridDim.x) {
    const int m0 = (t % nmt) * 64;
    const int k0 = (t / nmt) * 64;
    {
      const int m = m0 + (tid >> 2);
      const int c0 = k0 + (tid & 3) * 16;
      __bf16 v[16];
      if (m < M) {
#pragma unroll
        for (int e = 0; e < 16; ++e) {
          const int c = c0 + e;
          v[e] = (__bf16)((c < K) ? to_f32(src[(long)m * lda + c]) : 0.0f);
        }
      } else {
#pragma unroll
        for (int e = 0; e < 16; ++e) v[e] = (__bf16)0.0f;
      }
      *(bf16x8*)&tile[tid >> 2][(tid & 3) * 16] = *(const bf16x8*)&v[0];
      *(bf16x8*)&tile[tid >> 2][(tid & 3) * 16 + 8] = *(const bf16x8*)&v[8];
    }
    __syncthreads();
    {
      const int k = k0 + (tid >> 2);
      if (k < K) {
        const int r0 = (tid & 3) * 16;
        __bf16 v[16];
#pragma unroll
        for (int e = 0; e < 16; ++e) v[e] = tile[r0 + e][tid >> 2];
        __builtin_memcpy(&dst[(long)k * Mpad + m0 + r0], v,
                         16 * sizeof(__bf16));
      }
    }
    __syncthreads();
  }
}

template <typename T>
hipError_t transpose_to_bf16_t(
    hipStream_t stream,
    const T* src,
    long lda,
    int M,
    int K,
    __hip_bfloat16* dst,
    int Mpad) {
  int tiles = (Mpad / 64) * ceil_div(K, 64);
  int blocks = min(tiles, 2048);
  transpose_to_bf16_kernel<T><<<blocks, 256, 0, stream>>>(
      src, lda, M, K, (__bf16*)dst, Mpad);
  return hipGetLastError();
}

template hipError_t transpose_to_bf16_t<__hip_bfloat16>(hipStream_t, const __hip_bfloat16*, long, int, int, __hip_bfloat16*, int);

// ---------------------------------------------------------------- launchers

static int pick_splits(int M, int n_tiles) {
  // Fill the chip (256 CUs, 1-2 blocks of 4 waves each) without
  // multiplying the fp32-atomic epilogue traffic more than needed.
  int max_splits = max(1, M / (8 * BK));
  int want = max(1, 512 / max(1, n_tiles));
  return min(max_splits, want);
}

template <bool BF16, typename AccL>
static hipError_t launch_syrk(
    hipStream_t stream,
    float* out,
    int N,
    const AccL& acc,
    float beta,
    float coeff) {
  long n2 = (long)N * N;
  int threads = 256;
  scale_kernel<<<(n2 + threads - 1) / threads, threads, 0, stream>>>(
      out, n2, beta);
  constexpr int kslab = BF16 ? BKB : BK;
  const bool big = BF16 && N > 128;
  int bt = big ? BTB : BT;
  int nt = ceil_div(N, bt);
  int n_tiles = nt * (nt + 1) / 2;
  int splits = pick_splits(acc.M, n_tiles);
  int m_per_split = ceil_div(ceil_div(acc.M, splits), kslab) * kslab;
  splits = ceil_div(acc.M, m_per_split);
  dim3 grid(nt, nt, splits);
  if constexpr (BF16) {
    if (big) {
      syrk_kernel_bf16_big<AccL, AccL><<<grid, 256, 0, stream>>>(
          out, N, acc, acc, m_per_split, coeff, true);
    } else {
      syrk_kernel_bf16<AccL, AccL><<<grid, 256, 0, stream>>>(
          out, N, acc, acc, m_per_split, coeff, true);
    }
  } else {
    syrk_kernel<AccL, AccL><<<grid, 256, 0, stream>>>(
        out, N, acc, acc, m_per_split, coeff, true);
  }
  return hipGetLastError();
}

// bf16 inputs take the bf16-MFMA kernel (no precision loss: the data is
// already bf16); fp32/fp16 inputs take the exact-f32 MFMA kernel.
template <typename T>
constexpr bool use_bf16_mfma() {
  return std::is_same<T, __hip_bfloat16>::value;
}

template <typename T>
hipError_t cov_linear_t(
    hipStream_t stream,
    const T* a,
    long lda,
    int M,
    int K,
    bool bias,
    float* out,
    float beta,
    float coeff) {
  LinearAcc<T> acc{a, lda, M, K, K + (bias ? 1 : 0)};
  return launch_syrk<use_bf16_mfma<T>()>(stream, out, acc.Ncols, acc, beta, coeff);
}

template <typename T>
hipError_t cov_conv_g_t(
    hipStream_t stream,
    const T* g,
    int Nb,
    int C,
    int OH,
    int OW,
    float* out,
    float beta,
    float coeff) {
  ConvGradAcc<T> acc{g, C, OH, OW, Nb * OH * OW, C};
  return launch_syrk<use_bf16_mfma<T>()>(stream, out, acc.Ncols, acc, beta, coeff);
}

// SYRK over a pre-transposed [K][Mpad] bf16 image (im2col_tr_t /
// transpose_to_bf16_t output): every staging write is a contiguous 16 B
// LDS store regardless of the source layout.
hipError_t cov_trans_t(
    hipStream_t stream,
    const __hip_bfloat16* ws,
    int Mpad,
    int M,
    int K,
    bool bias,
    float* out,
    float beta,
    float coeff) {
  TransAcc<__hip_bfloat16> acc{ws, Mpad, M, K, K + (bias ? 1 : 0)};
  return launch_syrk<true>(stream, out, acc.Ncols, acc, beta, coeff);
}

// Explicit instantiations used by the binding.
template hipError_t cov_linear_t<float>(hipStream_t, const float*, long, int, int, bool, float*, float, float);
template hipError_t cov_linear_t<__hip_bfloat16>(hipStream_t, const __hip_bfloat16*, long, int, int, bool, float*, float, float);
template hipError_t cov_linear_t<__half>(hipStream_t, const __half*, long, int, int, bool, float*, float, float);
template hipError_t cov_conv_g_t<float>(hipStream_t, const float*, int, int, int, int, float*, float, float);
template hipError_t cov_conv_g_t<__hip_bfloat16>(hipStream_t, const __hip_bfloat16*, int, int, int, int, float*, float, float);
template hipError_t cov_conv_g_t<__half>(hipStream_t, const __half*, int, int, int, int, float*, float, float);

}  // namespace kfac
