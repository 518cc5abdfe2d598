// Batched blocked Cholesky: the hand-written diagonal-block step.
//
// rocSOLVER's potrf/potrs run at 2-4 TF on MI355X for K-FAC's batched
// shapes (measured, profiles/qdwh_bench.md) because the panel chain is
// latency-bound library code.  The blocked factorization here keeps the
// O(n^3) trailing updates in large batched GEMMs (driven from Python on
// the hipBLASLt xf32 path, ~305 TF at n=4608) and hand-writes only the
// O(n * nb^2) part rocSOLVER serializes: factorizing the NB x NB
// diagonal block and inverting its triangle, one workgroup per matrix,
// entirely LDS-resident.
//
// Kernel contract (row-major (B, n, n) fp32 stack, lower triangle):
//   for each matrix b: D = A[b, j:j+m, j:j+m]
//     D = L11 L11^T   (in-LDS unblocked Cholesky, diag floored at eps)
//     A[b, j:j+m, j:j+m] <- L11  (lower part; upper left untouched)
//     dinv[b]          <- L11^{-1}  (m x m lower, rest zeroed)
//
// The host loop then forms the panel L21 = A21 @ L11^{-T} and the
// trailing update A22 -= L21 L21^T as batched GEMMs, and the full
// triangular inverse / solves reuse dinv in GEMM-only block recurrences
// (kfac_amd/ops/blocked.py).

#include "common.h"

namespace kfac {

constexpr int CHOL_NB = 128;

// One workgroup (256 threads) per matrix.  LDS: the diagonal block and
// its inverse, 2 x 128 x 128 fp32 = 128 KiB (fits the 160 KiB LDS).
__global__ __launch_bounds__(256) void chol_diag_inv_kernel(
    float* __restrict__ a,      // (B, n, n)
    float* __restrict__ dinv,   // (B, NB, NB)
    int n,
    int j,
    int m,
    float eps) {
  __shared__ float d[CHOL_NB][CHOL_NB + 1];
  __shared__ float t[CHOL_NB][CHOL_NB + 1];
  const int tid = threadIdx.x;
  float* base = a + (size_t)blockIdx.x * n * n + (size_t)j * n + j;

  // load lower triangle of the block (upper mirrored for simplicity)
  for (int e = tid; e < m * m; e += 256) {
    const int r = e / m;
    const int c = e % m;
    d[r][c] = (c <= r) ? base[(size_t)r * n + c] : 0.0f;
  }
  __syncthreads();

  // unblocked Cholesky on the LDS tile
  for (int k = 0; k < m; ++k) {
    // pivot + column scale (the k-th column below the diagonal)
    if (tid == 0) {
      d[k][k] = sqrtf(fmaxf(d[k][k], eps));
    }
    __syncthreads();
    const float dk = d[k][k];
    for (int i = k + 1 + tid; i < m; i += 256) {
      d[i][k] /= dk;
    }
    __syncthreads();
    // trailing rank-1 update on the lower triangle
    const int rows = m - k - 1;
    for (int e = tid; e < rows * rows; e += 256) {
      const int r = k + 1 + e / rows;
      const int c = k + 1 + e % rows;
      if (c <= r) {
        d[r][c] -= d[r][k] * d[c][k];
      }
    }
    __syncthreads();
  }

  // write back L11 (lower)
  for (int e = tid; e < m * m; e += 256) {
    const int r = e / m;
    const int c = e % m;
    if (c <= r) {
      base[(size_t)r * n + c] = d[r][c];
    }
  }

  // in-LDS triangular inverse: one thread per column, forward
  // substitution; columns are independent.
  if (tid < m) {
    const int jc = tid;
    t[jc][jc] = 1.0f / d[jc][jc];
    for (int i = jc + 1; i < m; ++i) {
      float s = 0.0f;
      for (int k = jc; k < i; ++k) {
        s += d[i][k] * t[k][jc];
      }
      t[i][jc] = -s / d[i][i];
    }
  }
  __syncthreads();

  float* dv = dinv + (size_t)blockIdx.x * CHOL_NB * CHOL_NB;
  for (int e = tid; e < CHOL_NB * CHOL_NB; e += 256) {
    const int r = e / CHOL_NB;
    const int c = e % CHOL_NB;
    dv[e] = (r < m && c <= r) ? t[r][c] : 0.0f;
  }
}

hipError_t chol_diag_inv_f32(
    hipStream_t stream,
    float* a,
    float* dinv,
    int B,
    int n,
    int j,
    int m,
    float eps) {
  chol_diag_inv_kernel<<<B, 256, 0, stream>>>(a, dinv, n, j, m, eps);
  return hipGetLastError();
}

// ---- column gather/scatter for the warm block-Jacobi rounds ----
//
// The rotation rounds update pair COLUMNS of T (and Q) across the whole
// batch.  torch advanced indexing materializes p x rows x 2b int64
// index grids (~160 MB per round at n=4608) — these kernels move only
// the payload.  idx holds per-pair column indices (two contiguous
// b-runs, so the global accesses coalesce).

__global__ void gather_cols_kernel(
    const float* __restrict__ t,    // (B, rows, n) flat
    const long* __restrict__ mat,   // (p,)
    const long* __restrict__ idx,   // (p, m)
    float* __restrict__ out,        // (p, rows, m)
    int rows,
    int n,
    int m) {
  const long e = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)gridDim.x * blockDim.x;
  const long count = (long)rows * m;
  const long pi = blockIdx.y;
  for (long i = e; i < count; i += total) {
    const int r = (int)(i / m);
    const int c = (int)(i % m);
    out[(pi * rows + r) * (long)m + c] =
        t[(mat[pi] * rows + r) * (long)n + idx[pi * m + c]];
  }
}

__global__ void scatter_cols_kernel(
    float* __restrict__ t,
    const long* __restrict__ mat,
    const long* __restrict__ idx,
    const float* __restrict__ src,  // (p, rows, m)
    int rows,
    int n,
    int m) {
  const long e = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)gridDim.x * blockDim.x;
  const long count = (long)rows * m;
  const long pi = blockIdx.y;
  for (long i = e; i < count; i += total) {
    const int r = (int)(i / m);
    const int c = (int)(i % m);
    t[(mat[pi] * rows + r) * (long)n + idx[pi * m + c]] =
        src[(pi * rows + r) * (long)m + c];
  }
}

hipError_t gather_cols_f32(
    hipStream_t stream,
    const float* t,
    const long* mat,
    const long* idx,
    float* out,
    int p,
    int rows,
    int n,
    int m) {
  dim3 grid(64, p);
  gather_cols_kernel<<<grid, 256, 0, stream>>>(t, mat, idx, out, rows, n, m);
  return hipGetLastError();
}

hipError_t scatter_cols_f32(
    hipStream_t stream,
    float* t,
    const long* mat,
    const long* idx,
    const float* src,
    int p,
    int rows,
    int n,
    int m) {
  dim3 grid(64, p);
  scatter_cols_kernel<<<grid, 256, 0, stream>>>(t, mat, idx, src, rows, n, m);
  return hipGetLastError();
}

// ---- fused allreduce-bucket unpack (K13) ----
//
// The bucket pack is a single torch.cat launch; the unpack previously
// issued one copy kernel per tensor (~30-50 launches per factor step).
// One kernel scatters the whole flat buffer back through a descriptor
// table of (dst pointer, offset, numel).

struct BucketDesc {
  float* dst;
  long offset;
  long numel;
};

__global__ void bucket_unpack_kernel(
    const float* __restrict__ flat,
    const BucketDesc* __restrict__ desc,
    int n_tensors,
    long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    // binary search the tensor containing flat element i
    int lo = 0;
    int hi = n_tensors - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (desc[mid].offset <= i) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    desc[lo].dst[i - desc[lo].offset] = flat[i];
  }
}

hipError_t bucket_unpack_f32(
    hipStream_t stream,
    const float* flat,
    const void* desc,
    int n_tensors,
    long total) {
  const int threads = 256;
  const int blocks =
      (int)min((total + threads - 1) / threads, (long)8192);
  bucket_unpack_kernel<<<blocks, threads, 0, stream>>>(
      flat, (const BucketDesc*)desc, n_tensors, total);
  return hipGetLastError();
}

}  // namespace kfac
