// Batched symmetric eigensolver for small factors (K5 of SURVEY §2.4).
//
// Two-sided cyclic Jacobi, ONE WAVE per matrix (n <= 64), A and the
// eigenvector accumulator V LDS-resident (33 KB -> 4 blocks/CU). A
// single wave executes phases in lockstep, so the inter-phase barriers
// are single-wave s_barriers (near-free) instead of multi-wave
// rendezvous — the multi-wave variant measured barrier-latency-bound.
// Round-robin tournament pairs are precomputed into LDS each round so
// the update loops are free of div/mod.
//
// Why hand-written: rocSOLVER syevd is host-launch-bound for small n
// (~50k kernels per inverse phase) and its own syevj is 18x slower than
// syevd (profiles/eigh_strategies.md). Eigenvalues come out UNSORTED
// (K-FAC\'s Kronecker preconditioner is order-invariant) and clamped
// >= 0 as the eigen layer requires.

#include "common.h"

namespace kfac {

constexpr int JN = 64;   // matrix capacity
constexpr int JLD = JN + 1;

__global__ __launch_bounds__(64) void syevj_wave_kernel(
    const float* __restrict__ a_stack,  // [batch][n][n]
    float* __restrict__ w_out,          // [batch][n]
    float* __restrict__ v_out,          // [batch][n][n] (columns = vectors)
    int n,
    int max_sweeps,
    float tol) {
  __shared__ float As[JN][JLD];
  __shared__ float Vs[JN][JLD];
  __shared__ float cs[JN / 2];
  __shared__ float ss[JN / 2];
  __shared__ unsigned char ps[JN / 2];
  __shared__ unsigned char qs[JN / 2];
  __shared__ int converged;

  const int lane = threadIdx.x;
  const long mat = blockIdx.x;
  const float* A = a_stack + mat * (long)n * n;

  for (int idx = lane; idx < JN * JN; idx += 64) {
    int i = idx >> 6;
    int j = idx & 63;
    As[i][j] = (i < n && j < n) ? A[(long)i * n + j] : 0.0f;
    Vs[i][j] = (i == j) ? 1.0f : 0.0f;
  }
  __syncthreads();

  constexpr int m = JN;
  for (int sweep = 0; sweep < max_sweeps; ++sweep) {
    float off2 = 0.0f;
    float d2 = 0.0f;
    for (int idx = lane; idx < n * n; idx += 64) {
      int i = idx / n;
      int j = idx % n;
      float v = As[i][j];
      if (i != j) {
        off2 += v * v;
      } else {
        d2 += v * v;
      }
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      off2 += __shfl_down(off2, o, 64);
      d2 += __shfl_down(d2, o, 64);
    }
    if (lane == 0) {
      converged = (off2 <= tol * tol * (d2 + 1e-30f)) ? 1 : 0;
    }
    __syncthreads();
    if (converged) break;

    for (int r = 0; r < m - 1; ++r) {
      // phase 1: pair table + rotation params (lanes 0..31)
      if (lane < m / 2) {
        int j = lane;
        int p, q;
        if (j == 0) {
          p = m - 1;
          q = r;
        } else {
          p = (r + j) % (m - 1);
          q = (r - j + (m - 1)) % (m - 1);
        }
        int lo = min(p, q);
        int hi = max(p, q);
        ps[j] = (unsigned char)lo;
        qs[j] = (unsigned char)hi;
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
      // phase 2: row updates (rows partitioned across pairs)
#pragma unroll 4
      for (int idx = lane; idx < (m / 2) * m; idx += 64) {
        int j = idx >> 6;   // pair slot (m = 64)
        int k = idx & 63;
        int lo = ps[j];
        int hi = qs[j];
        float c = cs[j];
        float s = ss[j];
        float alo = As[lo][k];
        float ahi = As[hi][k];
        As[lo][k] = c * alo - s * ahi;
        As[hi][k] = s * alo + c * ahi;
      }
      __syncthreads();
      // phase 3: column updates of A and V (columns partitioned)
#pragma unroll 4
      for (int idx = lane; idx < (m / 2) * m; idx += 64) {
        int j = idx >> 6;
        int k = idx & 63;
        int lo = ps[j];
        int hi = qs[j];
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

  float* W = w_out + mat * (long)n;
  float* V = v_out + mat * (long)n * n;
  for (int i = lane; i < n; i += 64) {
    W[i] = fmaxf(As[i][i], 0.0f);
  }
  for (int idx = lane; idx < n * n; idx += 64) {
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
  if (n > JN) {
    return hipErrorInvalidValue;
  }
  syevj_wave_kernel<<<batch, 64, 0, stream>>>(
      a_stack, w_out, v_out, n, max_sweeps, tol);
  return hipGetLastError();
}

}  // namespace kfac
