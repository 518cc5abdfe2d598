// Batched symmetric eigensolver for small factors (K5 of SURVEY §2.4).
//
// Two-sided cyclic Jacobi, ONE WAVE per matrix (n <= 64), LDS-resident.
// Each phase reads one LDS buffer and writes the other (ping-pong): with
// a single array the compiler must assume every store may alias the next
// iteration's load and serializes the whole phase on lgkmcnt(0) — the
// aliased variant measured ~6.3 us per phase, latency-bound. Ping-pong
// makes all 64 load pairs of a phase independent so they pipeline.
// Round-robin tournament pairs give 32 independent rotations per round;
// rows are partitioned in the row phase and columns in the column phase,
// and since the 32 pairs cover all 64 indices both phases fully rewrite
// the destination buffer.
//
// Why hand-written: rocSOLVER syevd is host-launch-bound for small n
// (~50k kernels per inverse phase) and rocSOLVER's own syevj is 18x
// slower than syevd (profiles/eigh_strategies.md). Eigenvalues come out
// UNSORTED (the Kronecker preconditioner is order-invariant) and
// clamped >= 0 as the eigen layer requires.

#include "common.h"

namespace kfac {

constexpr int JN = 64;  // matrix capacity (padded with zeros)
constexpr int JLD = JN + 1;

__global__ __launch_bounds__(64) void syevj_wave_kernel(
    const float* __restrict__ a_stack,  // [batch][n][n]
    float* __restrict__ w_out,          // [batch][n]
    float* __restrict__ v_out,          // [batch][n][n] (columns = vectors)
    int n,
    int max_sweeps,
    float tol) {
  __shared__ float Abuf[2][JN][JLD];
  __shared__ float Vbuf[2][JN][JLD];
  __shared__ int converged;

  const int lane = threadIdx.x;
  const long mat = blockIdx.x;
  const float* A = a_stack + mat * (long)n * n;

  int acur = 0;  // which A buffer holds the current matrix
  int vcur = 0;
  for (int idx = lane; idx < JN * JN; idx += 64) {
    int i = idx >> 6;
    int j = idx & 63;
    Abuf[0][i][j] = (i < n && j < n) ? A[(long)i * n + j] : 0.0f;
    Vbuf[0][i][j] = (i == j) ? 1.0f : 0.0f;
  }
  __syncthreads();

  constexpr int m = JN;
  for (int sweep = 0; sweep < max_sweeps; ++sweep) {
    float off2 = 0.0f;
    float d2 = 0.0f;
    for (int i = 0; i < n; ++i) {
      if (lane < n) {
        float v = Abuf[acur][i][lane];
        if (i != lane) {
          off2 += v * v;
        } else {
          d2 += v * v;
        }
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
      auto Ain = Abuf[acur];
      auto Amid = Abuf[acur ^ 1];
      // phase 1: lane j < 32 computes its pair + rotation into REGISTERS;
      // phases 2/3 fetch them with __shfl (wave-synchronous, no LDS
      // round trip, no barrier).
      int mylo = 0;
      int myhi = 0;
      float myc = 1.0f;
      float mys = 0.0f;
      {
        int j = lane & 31;
        int p, q;
        if (j == 0) {
          p = m - 1;
          q = r;
        } else {
          p = (r + j) % (m - 1);
          q = (r - j + (m - 1)) % (m - 1);
        }
        mylo = min(p, q);
        myhi = max(p, q);
        float apq = Ain[mylo][myhi];
        if (fabsf(apq) > 1e-30f) {
          float tau = (Ain[myhi][myhi] - Ain[mylo][mylo]) / (2.0f * apq);
          float t = copysignf(1.0f, tau) /
                    (fabsf(tau) + sqrtf(1.0f + tau * tau));
          myc = rsqrtf(1.0f + t * t);
          mys = t * myc;
        }
      }
      __builtin_amdgcn_wave_barrier();
      // phase 2: row updates, Ain -> Amid (pairs cover every row)
#pragma unroll
      for (int j = 0; j < m / 2; ++j) {
        int lo = __shfl(mylo, j, 64);
        int hi = __shfl(myhi, j, 64);
        float c = __shfl(myc, j, 64);
        float s = __shfl(mys, j, 64);
        float alo = Ain[lo][lane];
        float ahi = Ain[hi][lane];
        Amid[lo][lane] = c * alo - s * ahi;
        Amid[hi][lane] = s * alo + c * ahi;
      }
      __syncthreads();
      // phase 3: column updates, Amid -> Ain, and V ping-pong
      auto Vin = Vbuf[vcur];
      auto Vout = Vbuf[vcur ^ 1];
#pragma unroll
      for (int j = 0; j < m / 2; ++j) {
        int lo = __shfl(mylo, j, 64);
        int hi = __shfl(myhi, j, 64);
        float c = __shfl(myc, j, 64);
        float s = __shfl(mys, j, 64);
        float alo = Amid[lane][lo];
        float ahi = Amid[lane][hi];
        Ain[lane][lo] = c * alo - s * ahi;
        Ain[lane][hi] = s * alo + c * ahi;
        float vlo = Vin[lane][lo];
        float vhi = Vin[lane][hi];
        Vout[lane][lo] = c * vlo - s * vhi;
        Vout[lane][hi] = s * vlo + c * vhi;
      }
      vcur ^= 1;
      __syncthreads();
    }
  }

  float* W = w_out + mat * (long)n;
  float* V = v_out + mat * (long)n * n;
  if (lane < n) {
    W[lane] = fmaxf(Abuf[acur][lane][lane], 0.0f);
  }
  for (int idx = lane; idx < n * n; idx += 64) {
    int i = idx / n;
    int j = idx % n;
    V[(long)i * n + j] = Vbuf[vcur][i][j];
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
