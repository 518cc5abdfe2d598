// Small-matrix MFMA GEMM with fused epilogues for the K-FAC
// precondition chain (K8/K9 of SURVEY §2.4):
//
//   eigen:   v1 = QG^T @ grad @ QA ; v2 = v1 * dGdA (or / (dg x da + damping))
//            out = QG @ v2 @ QA^T
//   inverse: out = G^-1 @ grad @ A^-1
//
// Factor edges are 64..~4k, so a 64x64-tile mfma_f32_16x16x4_f32 kernel
// with LDS staging covers the whole size range; the elementwise
// eigenvalue denominator is fused into the second GEMM's epilogue so the
// chain is exactly 4 kernel launches per layer with no intermediate
// elementwise pass over HBM.

#include "common.h"

namespace kfac {

constexpr int GBT = 128;
constexpr int GBK = 32;
// Unpadded rows: required for global_load_lds (lane-linear destination).
// Bank-conflict cost of the power-of-2 stride is minor here: the b32
// MFMA-loop reads go 2-way on half-wave groups and the trans-path writes
// 8-way on 4 instructions — tens of cycles against the ~4096-cycle f32
// MFMA issue time per K-slice.
constexpr int GLDS = GBT;

enum class Epilogue : int { NONE = 0, MUL = 1, DIV_OUTER = 2 };

typedef __attribute__((ext_vector_type(8))) __bf16 gemm_bf16x8;

// Split-precision operands: x = hi + lo with both parts bf16 keeps ~16
// mantissa bits per element, and 3 bf16 MFMAs (hi*hi + hi*lo + lo*hi)
// replace the 8 f32 MFMAs of a 32-deep K slice. f32 MFMA is the
// deliberately slow path on CDNA4 (155 TF vs 2.5 PF bf16), so this cuts
// the chain's MFMA issue cycles ~5x; the dropped lo*lo term is O(2^-32)
// relative and the end-to-end precondition error vs the fp64 reference
// stays ~1e-5 (tests/test_ops_gpu.py precond_* tolerances).
__device__ __forceinline__ void split_bf16(
    const float* v,
    gemm_bf16x8& hi,
    gemm_bf16x8& lo) {
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    __bf16 h = (__bf16)v[q];
    hi[q] = h;
    lo[q] = (__bf16)(v[q] - (float)h);  // exact f32 residual
  }
}

// LDS column swizzle at 4-float (16 B) granularity: distributes the
// trans-path writes (8 lanes sharing a column index across k-rows,
// measured 2.8e9 SQ_LDS_BANK_CONFLICT cycles on the stage-4 GEMM) over
// distinct banks, stays 16-byte aligned for float4 writes and glds, and
// is its own inverse so the glds SOURCE pre-swizzle and the read-side
// XOR are the same involution (guide rule 21).
__device__ __forceinline__ int lds_swz(int k, int i) {
  return i ^ (((k >> 2) & 7) << 2);
}

// Stage op(A)'s [k, rows i0..i0+GBT) slab into lds[k][i].
// The thread->element mapping follows the PHYSICAL memory order so loads
// coalesce either way: each thread moves 4 consecutive floats of the
// source (one 16-byte load in the interior).
__device__ __forceinline__ void stage_gemm(
    const float* __restrict__ src,
    int rows,      // logical i extent of op(src)
    int ks,        // logical k extent
    long ld,       // leading dim of the PHYSICAL matrix
    bool trans,    // false: physical[k][i] = src[k*ld+i] feeds lds[k][i]
                   // true:  physical[i][k] = src[i*ld+k] feeds lds[k][i]
    int k0,
    int i0,
    float (*lds)[GLDS],
    int tid) {
  if (!trans && (ld & 3) == 0 && i0 + GBT <= rows && k0 + GBK <= ks) {
    // Full interior tile, 16-byte-aligned rows: async global->LDS DMA.
    // Each wave chunk is 2 LDS rows (64 lanes x 16 B = 1 KiB); the
    // per-lane SOURCE address mirrors the lane-linear LDS layout.
    const int wv = tid >> 6;
    const int l = tid & 63;
    // 1 KiB chunks (2 rows): GBK*GBT floats / 256 per chunk, 4 waves
    constexpr int kChunks = (GBK * GBT) / 256;        // 16
    constexpr int kPerWave = kChunks / 4;             // 4
#pragma unroll
    for (int e = 0; e < kPerWave; ++e) {
      const int ci = wv * kPerWave + e;
      const int krow = 2 * ci + (l >> 5);
      const float* gp =
          src + (long)(k0 + krow) * ld + (i0 + lds_swz(krow, (l & 31) * 4));
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gp,
          (__attribute__((address_space(3))) void*)&lds[2 * ci][0],
          16,
          0,
          0);
    }
    return;
  }
#pragma unroll
  for (int e = 0; e < (GBK * GBT) / (256 * 4); ++e) {
    const int c = tid + e * 256;
    if (trans) {
      // contiguous along k in memory: 8 threads cover one 32-k row
      const int k = (c & 7) * 4;
      const int i = c >> 3;
      const int gi = i0 + i;
      const long base = (long)gi * ld + (k0 + k);
      float v[4];
      if (gi < rows && k0 + k + 3 < ks) {
        __builtin_memcpy(v, src + base, 16);
      } else {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          v[q] = (gi < rows && k0 + k + q < ks) ? src[base + q] : 0.0f;
        }
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) lds[k + q][lds_swz(k + q, i)] = v[q];
    } else {
      // contiguous along i in memory: 32 threads cover one 128-i row
      const int i = (c & 31) * 4;
      const int k = c >> 5;
      const int gk = k0 + k;
      const long base = (long)gk * ld + (i0 + i);
      float v[4];
      if (gk < ks && i0 + i + 3 < rows) {
        __builtin_memcpy(v, src + base, 16);
      } else {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          v[q] = (gk < ks && i0 + i + q < rows) ? src[base + q] : 0.0f;
        }
      }
      *(float4*)&lds[k][lds_swz(k, i)] = make_float4(v[0], v[1], v[2], v[3]);
    }
  }
}

// Register staging for the pipelined split path: each thread owns 16
// elements (4 float4) of one matrix's 32x128 slab. Global loads for
// slice t+1 issue right after slice t's LDS writes, so their HBM
// latency hides behind slice t's MFMAs (guide T14: one register set,
// write after the barrier, re-issue immediately) — the glds path's
// issue cost (~60-185 cyc per 1 KiB piece) is no longer small against
// the bf16x3 MFMA budget (~670 cyc/slice vs f32's ~3600).
struct StageRegs {
  float4 v[4];
};

__device__ __forceinline__ void stage_load(
    const float* __restrict__ src,
    int rows,
    int ks,
    long ld,
    bool trans,
    int k0,
    int i0,
    int tid,
    StageRegs& r) {
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    const int c = tid + e * 256;
    float v[4];
    if (trans) {
      const int k = (c & 7) * 4;
      const int i = c >> 3;
      const int gi = i0 + i;
      const long base = (long)gi * ld + (k0 + k);
      if (gi < rows && k0 + k + 3 < ks) {
        __builtin_memcpy(v, src + base, 16);
      } else {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          v[q] = (gi < rows && k0 + k + q < ks) ? src[base + q] : 0.0f;
        }
      }
    } else {
      const int i = (c & 31) * 4;
      const int k = c >> 5;
      const int gk = k0 + k;
      const long base = (long)gk * ld + (i0 + i);
      if (gk < ks && i0 + i + 3 < rows) {
        __builtin_memcpy(v, src + base, 16);
      } else {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          v[q] = (gk < ks && i0 + i + q < rows) ? src[base + q] : 0.0f;
        }
      }
    }
    r.v[e] = make_float4(v[0], v[1], v[2], v[3]);
  }
}

__device__ __forceinline__ void stage_store(
    const StageRegs& r,
    bool trans,
    int tid,
    float (*lds)[GLDS]) {
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    const int c = tid + e * 256;
    if (trans) {
      const int k = (c & 7) * 4;
      const int i = c >> 3;
      const float* v = (const float*)&r.v[e];
#pragma unroll
      for (int q = 0; q < 4; ++q) lds[k + q][lds_swz(k + q, i)] = v[q];
    } else {
      const int i = (c & 31) * 4;
      const int k = c >> 5;
      *(float4*)&lds[k][lds_swz(k, i)] = r.v[e];
    }
  }
}

// One 128x128 output tile of C = op(A) @ op(B) (+ epilogue).
// 4 waves, each owning a 64x64 quadrant as 4x4 16x16 MFMA fragments
// (64 f32 accumulators/lane); per 32-deep K slice each wave issues 128
// mfma_f32_16x16x4_f32 against 16 LDS reads per substep — the MFMA:LDS
// ratio that the 64x64 structure lacked (guide §5 step-2 ladder).
// The SPLIT (bf16x3) instantiation instead runs the register-staged
// pipeline above with 48 bf16 MFMAs per slice.
template <Epilogue EPI, bool SPLIT>
__device__ __forceinline__ void gemm_tile_body(
    float* __restrict__ c,
    const float* __restrict__ a,
    const float* __restrict__ b,
    int M,
    int N,
    int K,
    bool ta,  // op(A)[m][k]; ta=false: A[m][k]; ta=true: A[k][m]
    bool tb,  // op(B)[k][n]; tb=false: B[k][n]; tb=true: B[n][k]
    long lda,  // physical leading dim of a (may exceed the logical dim
    long ldb,  // for padded scratch buffers -> keeps the glds path alive)
    long ldc,
    const float* __restrict__ e1,  // MUL: dgda [M x N]; DIV_OUTER: dg [M]
    const float* __restrict__ e2,  // DIV_OUTER: da [N]
    float damping,
    int i0,
    int j0,
    float (*lds_a)[GLDS],
    float (*lds_b)[GLDS]) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  f32x4 acc[4][4] = {};

  if constexpr (SPLIT) {
    // Register-staged pipeline with DOUBLE-BUFFERED LDS: slice t+1's
    // staging writes land in the other slab while slice t's MFMAs read
    // — ONE barrier per slice instead of two (the stage kernels
    // measured issue/barrier-bound at ~15% MFMA-pipe occupancy,
    // profiles/pmc_r2.txt).  Slabs are lds_a + p*GBK rows.
    StageRegs ra, rb;
    stage_load(a, M, K, lda, !ta, 0, i0, tid, ra);
    stage_load(b, N, K, ldb, tb, 0, j0, tid, rb);
    stage_store(ra, !ta, tid, lds_a);
    stage_store(rb, tb, tid, lds_b);
    if (GBK < K) {
      stage_load(a, M, K, lda, !ta, GBK, i0, tid, ra);
      stage_load(b, N, K, ldb, tb, GBK, j0, tid, rb);
    }
    __syncthreads();  // slab 0 ready
    int p = 0;
    for (int k0 = 0; k0 < K; k0 += GBK) {
      // stage slice t+1 into the OTHER slab (no readers there)
      if (k0 + GBK < K) {
        stage_store(ra, !ta, tid, lds_a + (p ^ 1) * GBK);
        stage_store(rb, tb, tid, lds_b + (p ^ 1) * GBK);
        if (k0 + 2 * GBK < K) {
          stage_load(a, M, K, lda, !ta, k0 + 2 * GBK, i0, tid, ra);
          stage_load(b, N, K, ldb, tb, k0 + 2 * GBK, j0, tid, rb);
        }
      }
      // One 16x16x32 bf16 MFMA triple covers the whole 32-deep slice:
      // lane l supplies elements k = 8*(l>>4)..+7 of column (l&15).
      float(*cur_a)[GLDS] = lds_a + p * GBK;
      float(*cur_b)[GLDS] = lds_b + p * GBK;
      const int kbase = (lane >> 4) * 8;
      gemm_bf16x8 ah[4], al[4], bh[4], bl[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int ca = wr * 64 + f * 16 + (lane & 15);
        const int cb = wc * 64 + f * 16 + (lane & 15);
        float va[8], vb[8];
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          va[q] = cur_a[kbase + q][lds_swz(kbase + q, ca)];
          vb[q] = cur_b[kbase + q][lds_swz(kbase + q, cb)];
        }
        split_bf16(va, ah[f], al[f]);
        split_bf16(vb, bh[f], bl[f]);
      }
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          // cross terms first so the large hi*hi lands last in the chain
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ah[fi], bl[fj], acc[fi][fj], 0, 0, 0);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              al[fi], bh[fj], acc[fi][fj], 0, 0, 0);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ah[fi], bh[fj], acc[fi][fj], 0, 0, 0);
        }
      }
      __syncthreads();  // this slice's reads AND next slab's writes done
      p ^= 1;
    }
  } else {
    for (int k0 = 0; k0 < K; k0 += GBK) {
      stage_gemm(a, M, K, lda, !ta, k0, i0, lds_a, tid);
      stage_gemm(b, N, K, ldb, tb, k0, j0, lds_b, tid);
      __syncthreads();
#pragma unroll
      for (int kk = 0; kk < GBK; kk += 4) {
        const int krow = kk + (lane >> 4);
        float av[4];
        float bv[4];
#pragma unroll
        for (int f = 0; f < 4; ++f) {
          av[f] = lds_a[krow][lds_swz(krow, wr * 64 + f * 16 + (lane & 15))];
          bv[f] = lds_b[krow][lds_swz(krow, wc * 64 + f * 16 + (lane & 15))];
        }
#pragma unroll
        for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
          for (int fj = 0; fj < 4; ++fj) {
            acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                av[fi], bv[fj], acc[fi][fj], 0, 0, 0);
          }
        }
      }
      __syncthreads();
    }
  }

#pragma unroll
  for (int fi = 0; fi < 4; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 4; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i0 + wr * 64 + fi * 16 + (lane >> 4) * 4 + r;
        int col = j0 + wc * 64 + fj * 16 + (lane & 15);
        if (row < M && col < N) {
          float v = acc[fi][fj][r];
          if constexpr (EPI == Epilogue::MUL) {
            v *= e1[(long)row * N + col];
          } else if constexpr (EPI == Epilogue::DIV_OUTER) {
            v /= (e1[row] * e2[col] + damping);
          }
          c[(long)row * ldc + col] = v;
        }
      }
    }
  }
}

template <Epilogue EPI, bool SPLIT>
__global__ __launch_bounds__(256) void gemm_kernel(
    float* __restrict__ c,
    const float* __restrict__ a,
    const float* __restrict__ b,
    int M,
    int N,
    int K,
    bool ta,
    bool tb,
    const float* __restrict__ e1,
    const float* __restrict__ e2,
    float damping) {
  // 2x GBK rows: the SPLIT path ping-pongs two slabs (single-barrier
  // pipeline); the f32 path uses only the first GBK rows.
  __shared__ float lds_a[2 * GBK][GLDS];
  __shared__ float lds_b[2 * GBK][GLDS];
  gemm_tile_body<EPI, SPLIT>(
      c, a, b, M, N, K, ta, tb, ta ? (long)M : (long)K,
      tb ? (long)K : (long)N, (long)N, e1, e2, damping,
      (int)blockIdx.x * GBT, (int)blockIdx.y * GBT, lds_a, lds_b);
}

// ---------------------------------------------------- grouped precondition
//
// The whole eigen precondition chain for EVERY layer in 4 kernel launches
// (the reference launches ~8 torch ops per layer, eigen.py:374-385; the
// per-layer v1 loop is launch-bound on 50+ layer models). Stages:
//   1: s1  = QG^T @ grad
//   2: s2  = (s1 @ QA) * dgda
//   3: s1  = QG @ s2
//   4: out = s1 @ QA^T
// Every stage's output is (m, n) so one tile table serves all stages.

struct PrecondDesc {
  long m, n;
  const float* grad;
  const float* qa;
  const float* qg;
  const float* dgda;
  float* s1;
  float* s2;
  float* out;
  long tile_off;  // first global tile index of this layer
  // fully-fused path (gather/scatter directly on module gradients)
  float* wgrad;   // weight grad storage [m][n - has_bias] (fp32)
  float* bgrad;   // bias grad [m] or nullptr
  long spad;      // physical row stride of grad/s1/s2/out buffers (>= n,
                  // multiple of 4 so staging keeps the glds fast path)
};

template <int STAGE>
__global__ __launch_bounds__(256) void grouped_precond_kernel(
    const PrecondDesc* __restrict__ desc,
    int n_layers) {
  const int tile = blockIdx.x;
  int l = 0;
  while (l + 1 < n_layers && tile >= (int)desc[l + 1].tile_off) ++l;
  const PrecondDesc d = desc[l];
  const int m = (int)d.m;
  const int n = (int)d.n;
  const int local = tile - (int)d.tile_off;
  const int ntj = ceil_div(n, GBT);
  const int i0 = (local / ntj) * GBT;
  const int j0 = (local % ntj) * GBT;

  __shared__ float lds_a[2 * GBK][GLDS];
  __shared__ float lds_b[2 * GBK][GLDS];

  const long sp = d.spad;
  if constexpr (STAGE == 1) {
    gemm_tile_body<Epilogue::NONE, true>(
        d.s1, d.qg, d.grad, m, n, m, true, false, (long)m, sp, sp, nullptr,
        nullptr, 0.f, i0, j0, lds_a, lds_b);
  } else if constexpr (STAGE == 2) {
    gemm_tile_body<Epilogue::MUL, true>(
        d.s2, d.s1, d.qa, m, n, n, false, false, sp, (long)n, sp, d.dgda,
        nullptr, 0.f, i0, j0, lds_a, lds_b);
  } else if constexpr (STAGE == 3) {
    gemm_tile_body<Epilogue::NONE, true>(
        d.s1, d.qg, d.s2, m, n, m, false, false, (long)m, sp, sp, nullptr,
        nullptr, 0.f, i0, j0, lds_a, lds_b);
  } else {
    gemm_tile_body<Epilogue::NONE, true>(
        d.out, d.s1, d.qa, m, n, n, false, true, sp, (long)n, sp, nullptr,
        nullptr, 0.f, i0, j0, lds_a, lds_b);
  }
}

// Gather module grads into the flat fp32 matrix (bias as last column),
// or scatter the preconditioned result back scaled by *scale.
template <bool SCATTER>
__global__ __launch_bounds__(256) void grouped_grad_copy_kernel(
    const PrecondDesc* __restrict__ desc,
    int n_layers,
    const float* __restrict__ scale) {
  const int tile = blockIdx.x;
  int l = 0;
  while (l + 1 < n_layers && tile >= (int)desc[l + 1].tile_off) ++l;
  const PrecondDesc d = desc[l];
  const int m = (int)d.m;
  const int n = (int)d.n;
  const int local = tile - (int)d.tile_off;
  const int ntj = ceil_div(n, GBT);
  const int i0 = (local / ntj) * GBT;
  const int j0 = (local % ntj) * GBT;
  const int n_w = (d.bgrad != nullptr) ? n - 1 : n;
  const float sc = SCATTER ? *scale : 0.0f;
  for (int e = threadIdx.x; e < GBT * GBT; e += 256) {
    const int row = i0 + e / GBT;
    const int col = j0 + e % GBT;
    if (row >= m || col >= n) continue;
    if constexpr (SCATTER) {
      const float v = d.out[(long)row * d.spad + col] * sc;
      if (col < n_w) {
        d.wgrad[(long)row * n_w + col] = v;
      } else {
        d.bgrad[row] = v;
      }
    } else {
      float v;
      if (col < n_w) {
        v = d.wgrad[(long)row * n_w + col];
      } else {
        v = d.bgrad[row];
      }
      // grad buffer doubles as stage-1 B operand
      ((float*)d.grad)[(long)row * d.spad + col] = v;
    }
  }
}

// Grouped kl-clip dot product against the MODULE gradients: each tile
// accumulates sum(precon .* [wgrad|bgrad]) into *accum.  Used by the
// broadcast-composing apply path (HYBRID/MEM-OPT), where the precon
// grads already exist per layer (grouped chain or received broadcast)
// and only clip+scale+write-back remain to be fused.
__global__ __launch_bounds__(256) void grouped_dot_kernel(
    const PrecondDesc* __restrict__ desc,
    int n_layers,
    float* __restrict__ accum) {
  const int tile = blockIdx.x;
  int l = 0;
  while (l + 1 < n_layers && tile >= (int)desc[l + 1].tile_off) ++l;
  const PrecondDesc d = desc[l];
  const int m = (int)d.m;
  const int n = (int)d.n;
  const int local = tile - (int)d.tile_off;
  const int ntj = ceil_div(n, GBT);
  const int i0 = (local / ntj) * GBT;
  const int j0 = (local % ntj) * GBT;
  const int n_w = (d.bgrad != nullptr) ? n - 1 : n;
  float s = 0.0f;
  for (int e = threadIdx.x; e < GBT * GBT; e += 256) {
    const int row = i0 + e / GBT;
    const int col = j0 + e % GBT;
    if (row >= m || col >= n) continue;
    const float p = d.out[(long)row * d.spad + col];
    const float g =
        (col < n_w) ? d.wgrad[(long)row * n_w + col] : d.bgrad[row];
    s += p * g;
  }
  __shared__ float wsum[4];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_down(s, off, 64);
  }
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) wsum[wave] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    atomicAdd(accum, wsum[0] + wsum[1] + wsum[2] + wsum[3]);
  }
}

hipError_t precond_apply_only_f32(
    hipStream_t stream,
    const void* desc_dev,
    int n_layers,
    int total_tiles,
    float* accum,   // PRE-ACCUMULATED with any per-layer contributions
    float* scale,
    float kl_clip,
    float lr);

// scale = min(1, sqrt(kl_clip / |accum * lr^2|)); kl_clip <= 0 -> 1.
__global__ void grad_scale_kernel(
    float* __restrict__ scale,
    const float* __restrict__ accum,
    float kl_clip,
    float lr) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    float s = 1.0f;
    if (kl_clip > 0.0f) {
      float vg = fabsf(*accum * lr * lr);
      if (vg > 0.0f) {
        s = fminf(1.0f, sqrtf(kl_clip / vg));
      }
    }
    *scale = s;
  }
}

__global__ void zero_kernel(float* p) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *p = 0.0f;
}

template <typename T>
hipError_t kl_clip_accum_t(hipStream_t, float*, const T*, const T*, long);

hipError_t precond_apply_grouped_f32(
    hipStream_t stream,
    const void* desc_dev,
    int n_layers,
    int total_tiles,
    float* gbuf,     // flat gathered grads (== concatenated desc.grad)
    float* outbuf,   // flat outputs (== concatenated desc.out)
    long total_elems,
    float* accum,    // 1-elem workspace
    float* scale,    // 1-elem output
    float kl_clip,
    float lr) {
  auto desc = (const PrecondDesc*)desc_dev;
  grouped_grad_copy_kernel<false>
      <<<total_tiles, 256, 0, stream>>>(desc, n_layers, nullptr);
  grouped_precond_kernel<1><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<2><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<3><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<4><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  // accum arrives zeroed (or PRE-accumulated with contributions from
  // layers preconditioned outside this call) from the host binding.
  KFAC_HIP_CHECK(
      kl_clip_accum_t<float>(stream, accum, outbuf, gbuf, total_elems));
  grad_scale_kernel<<<1, 1, 0, stream>>>(scale, accum, kl_clip, lr);
  grouped_grad_copy_kernel<true>
      <<<total_tiles, 256, 0, stream>>>(desc, n_layers, scale);
  return hipGetLastError();
}

hipError_t precond_apply_only_f32(
    hipStream_t stream,
    const void* desc_dev,
    int n_layers,
    int total_tiles,
    float* accum,
    float* scale,
    float kl_clip,
    float lr) {
  auto desc = (const PrecondDesc*)desc_dev;
  grouped_dot_kernel<<<total_tiles, 256, 0, stream>>>(
      desc, n_layers, accum);
  grad_scale_kernel<<<1, 1, 0, stream>>>(scale, accum, kl_clip, lr);
  grouped_grad_copy_kernel<true>
      <<<total_tiles, 256, 0, stream>>>(desc, n_layers, scale);
  return hipGetLastError();
}

hipError_t precond_grouped_f32(
    hipStream_t stream,
    const void* desc_dev,
    int n_layers,
    int total_tiles) {
  auto desc = (const PrecondDesc*)desc_dev;
  grouped_precond_kernel<1><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<2><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<3><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  grouped_precond_kernel<4><<<total_tiles, 256, 0, stream>>>(desc, n_layers);
  return hipGetLastError();
}

hipError_t gemm_f32(
    hipStream_t stream,
    float* c,
    const float* a,
    const float* b,
    int M,
    int N,
    int K,
    bool ta,
    bool tb,
    int epilogue,
    const float* e1,
    const float* e2,
    float damping,
    bool split) {
  dim3 grid(ceil_div(M, GBT), ceil_div(N, GBT));
  switch (static_cast<Epilogue>(epilogue)) {
    case Epilogue::NONE:
      if (split) {
        gemm_kernel<Epilogue::NONE, true><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, nullptr, nullptr, 0.0f);
      } else {
        gemm_kernel<Epilogue::NONE, false><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, nullptr, nullptr, 0.0f);
      }
      break;
    case Epilogue::MUL:
      if (split) {
        gemm_kernel<Epilogue::MUL, true><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, e1, nullptr, 0.0f);
      } else {
        gemm_kernel<Epilogue::MUL, false><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, e1, nullptr, 0.0f);
      }
      break;
    case Epilogue::DIV_OUTER:
      if (split) {
        gemm_kernel<Epilogue::DIV_OUTER, true><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, e1, e2, damping);
      } else {
        gemm_kernel<Epilogue::DIV_OUTER, false><<<grid, 256, 0, stream>>>(
            c, a, b, M, N, K, ta, tb, e1, e2, damping);
      }
      break;
  }
  return hipGetLastError();
}

// ------------------------------------------------------- kl-clip reduction

// accum += sum(precon * grad) without any host round trip (K10).
template <typename T>
__global__ void kl_clip_kernel(
    float* __restrict__ accum,
    const T* __restrict__ precon,
    const T* __restrict__ grad,
    long n) {
  __shared__ float warp_sums[4];
  float s = 0.0f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    s += to_f32(precon[i]) * to_f32(grad[i]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_down(s, off, 64);
  }
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wave] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.0f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += warp_sums[w];
    atomicAdd(accum, t);
  }
}

template <typename T>
hipError_t kl_clip_accum_t(
    hipStream_t stream,
    float* accum,
    const T* precon,
    const T* grad,
    long n) {
  int threads = 256;
  int blocks = (int)min((n + threads - 1) / threads, (long)1024);
  blocks = max(blocks, 1);
  kl_clip_kernel<T><<<blocks, threads, 0, stream>>>(accum, precon, grad, n);
  return hipGetLastError();
}

template hipError_t kl_clip_accum_t<float>(hipStream_t, float*, const float*, const float*, long);
template hipError_t kl_clip_accum_t<__hip_bfloat16>(hipStream_t, float*, const __hip_bfloat16*, const __hip_bfloat16*, long);
template hipError_t kl_clip_accum_t<__half>(hipStream_t, float*, const __half*, const __half*, long);

// ------------------------------------------------------- triu pack/unpack

// Symmetric wire format (K12): packed index of (i,j), j>=i is
// i*n - i*(i-1)/2 + (j-i).
__global__ void triu_pack_kernel(
    float* __restrict__ packed,
    const float* __restrict__ x,
    int n) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long total = (long)n * n;
  if (idx >= total) return;
  int i = (int)(idx / n);
  int j = (int)(idx % n);
  if (j < i) return;
  long p = (long)i * n - (long)i * (i - 1) / 2 + (j - i);
  packed[p] = x[idx];
}

__global__ void triu_unpack_kernel(
    float* __restrict__ x,
    const float* __restrict__ packed,
    int n) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long total = (long)n * n;
  if (idx >= total) return;
  int i = (int)(idx / n);
  int j = (int)(idx % n);
  int r = min(i, j);
  int c = max(i, j);
  long p = (long)r * n - (long)r * (r - 1) / 2 + (c - r);
  x[idx] = packed[p];
}

hipError_t triu_pack_f32(hipStream_t stream, float* packed, const float* x, int n) {
  long total = (long)n * n;
  int threads = 256;
  triu_pack_kernel<<<(total + threads - 1) / threads, threads, 0, stream>>>(
      packed, x, n);
  return hipGetLastError();
}

hipError_t triu_unpack_f32(hipStream_t stream, float* x, const float* packed, int n) {
  long total = (long)n * n;
  int threads = 256;
  triu_unpack_kernel<<<(total + threads - 1) / threads, threads, 0, stream>>>(
      x, packed, n);
  return hipGetLastError();
}

}  // namespace kfac
