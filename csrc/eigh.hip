// Batched symmetric eigensolver for small factors (K5 of SURVEY §2.4).
//
// Two-sided cyclic Jacobi, one workgroup per matrix, A and the
// eigenvector accumulator V both LDS-resident (NMAX=64: 33 KB -> 4
// blocks/CU; NMAX=128: 132 KB -> 1 block/CU). Round-robin tournament
// ordering gives NMAX/2 independent rotations per round; each round is
// three barrier phases (rotation params -> row updates -> col+V updates)
// so no two rotations ever touch the same element in a phase.
//
// Why hand-written: rocSOLVER's syevd costs ~50k host-launched kernels
// per batch (launch-bound for small n) and its syevj is 18x slower than
// syevd (profiles/eigh_strategies.md); this kernel is ONE launch per
// group and leaves the eigenvalues unsorted (K-FAC's Kronecker
// preconditioner is order-invariant as long as d and Q columns match).

#include "common.h"

namespace kfac {

// round-robin pair schedule: round r (0..m-2), slot j (0..m/2-1)
__device__ __forceinline__ void rr_pair(int m, int r, int j, int* p, int* q) {
  if (j == 0) {
    *p = m - 1;
    *q = r;
  } else {
    *p = (r + j) % (m - 1);
    *q = (r - j + (m - 1)) % (m - 1);
  }
}

template <int NMAX>
__global__ __launch_bounds__(256) void syevj_small_kernel(
    const float* __restrict__ a_stack,  // [batch][n][n]
    float* __restrict__ w_out,          // [batch][n]
    float* __restrict__ v_out,          // [batch][n][n] (columns = vectors)
    int n,
    int max_sweeps,
    float tol) {
  constexpr int LD = NMAX + 1;
  __shared__ float As[NMAX][LD];
  __shared__ float Vs[NMAX][LD];
  __shared__ float cs[NMAX / 2];
  __shared__ float ss[NMAX / 2];
  __shared__ float red[4];
  __shared__ int converged;

  const int tid = threadIdx.x;
  const long mat = blockIdx.x;
  const float* A = a_stack + mat * (long)n * n;

  // load (zero-pad to NMAX; padding stays diagonal so it never mixes)
  for (int idx = tid; idx < NMAX * NMAX; idx += 256) {
    int i = idx / NMAX;
    int j = idx % NMAX;
    As[i][j] = (i < n && j < n) ? A[(long)i * n + j] : 0.0f;
    Vs[i][j] = (i == j) ? 1.0f : 0.0f;
  }
  __syncthreads();

  const int m = NMAX;
  for (int sweep = 0; sweep < max_sweeps; ++sweep) {
    // convergence: off(A)^2 <= tol^2 * diag(A)^2
    float off2 = 0.0f;
    for (int idx = tid; idx < n * n; idx += 256) {
      int i = idx / n;
      int j = idx % n;
      float v = As[i][j];
      if (i != j) off2 += v * v;
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      off2 += __shfl_down(off2, o, 64);
    }
    if ((tid & 63) == 0) {
      red[tid >> 6] = off2;
    }
    __syncthreads();
    if (tid == 0) {
      float o_total = red[0] + red[1] + red[2] + red[3];
      float d2 = 0.0f;
      for (int i = 0; i < n; ++i) d2 += As[i][i] * As[i][i];
      converged = (o_total <= tol * tol * (d2 + 1e-30f)) ? 1 : 0;
    }
    __syncthreads();
    if (converged) break;
    __syncthreads();

    for (int r = 0; r < m - 1; ++r) {
      // phase 1: rotation parameters from the untouched matrix
      for (int j = tid; j < m / 2; j += 256) {
        int p, q;
        rr_pair(m, r, j, &p, &q);
        int lo = min(p, q);
        int hi = max(p, q);
        float apq = As[lo][hi];
        float c = 1.0f;
        float s = 0.0f;
        if (fabsf(apq) > 1e-30f) {
          float tau = (As[hi][hi] - As[lo][lo]) / (2.0f * apq);
          float t = copysignf(1.0f, tau) /
                    (fabsf(tau) + sqrtf(1.0f + tau * tau));
          c = rsqrtf(1.0f + t * t);
          s = t * c;
        }
        cs[j] = c;
        ss[j] = s;
      }
      __syncthreads();
      // phase 2: row updates (rows are partitioned across pairs)
      for (int idx = tid; idx < (m / 2) * m; idx += 256) {
        int j = idx / m;
        int k = idx % m;
        int p, q;
        rr_pair(m, r, j, &p, &q);
        int lo = min(p, q);
        int hi = max(p, q);
        float c = cs[j];
        float s = ss[j];
        float alo = As[lo][k];
        float ahi = As[hi][k];
        As[lo][k] = c * alo - s * ahi;
        As[hi][k] = s * alo + c * ahi;
      }
      __syncthreads();
      // phase 3: column updates of A and V (columns partitioned)
      for (int idx = tid; idx < (m / 2) * m; idx += 256) {
        int j = idx / m;
        int k = idx % m;
        int p, q;
        rr_pair(m, r, j, &p, &q);
        int lo = min(p, q);
        int hi = max(p, q);
        float c = cs[j];
        float s = ss[j];
        float alo = As[k][lo];
        float ahi = As[k][hi];
        As[k][lo] = c * alo - s * ahi;
        As[k][hi] = s * alo + c * ahi;
        float vlo = Vs[k][lo];
        float vhi = Vs[k][hi];
        Vs[k][lo] = c * vlo - s * vhi;
        Vs[k][hi] = s * vlo + c * vhi;
      }
      __syncthreads();
    }
  }

  // eigenvalues (diag, clamped >= 0 as K-FAC requires) and vectors
  float* W = w_out + mat * (long)n;
  float* V = v_out + mat * (long)n * n;
  for (int i = tid; i < n; i += 256) {
    W[i] = fmaxf(As[i][i], 0.0f);
  }
  for (int idx = tid; idx < n * n; idx += 256) {
    int i = idx / n;
    int j = idx % n;
    V[(long)i * n + j] = Vs[i][j];
  }
}

hipError_t syevj_small_f32(
    hipStream_t stream,
    const float* a_stack,
    float* w_out,
    float* v_out,
    int n,
    int batch,
    int max_sweeps,
    float tol) {
  if (n <= 64) {
    syevj_small_kernel<64><<<batch, 256, 0, stream>>>(
        a_stack, w_out, v_out, n, max_sweeps, tol);
  } else if (n <= 128) {
    syevj_small_kernel<128><<<batch, 256, 0, stream>>>(
        a_stack, w_out, v_out, n, max_sweeps, tol);
  } else {
    return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

}  // namespace kfac
