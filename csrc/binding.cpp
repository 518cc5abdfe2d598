// Python binding for the kfac_amd HIP extension (_kfaccore).
//
// Host-side glue only: tensor checks, dtype dispatch, temp allocation,
// launches on the current HIP stream. Device code lives in syrk.hip /
// gemm.hip (pure HIP, gfx950).

#include <torch/extension.h>

#include <map>
#include <mutex>
#include <tuple>

#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>

namespace kfac {

template <typename T>
hipError_t cov_linear_t(hipStream_t, const T*, long, int, int, bool, float*, float, float);
template <typename T>
hipError_t im2col_t(hipStream_t, const T*, T*, int, int, int, int, int, int, int, int, int, int, int);
template <typename T>
hipError_t cov_conv_g_t(hipStream_t, const T*, int, int, int, int, float*, float, float);
template <typename T>
hipError_t im2col_tr_t(hipStream_t, const T*, T*, int, int, int, int, int, int, int, int, int, int, int);
template <typename T>
hipError_t transpose_to_bf16_t(hipStream_t, const T*, long, int, int, __hip_bfloat16*, int);
hipError_t cov_trans_t(hipStream_t, const __hip_bfloat16*, int, int, int, bool, float*, float, float);
hipError_t gemm_f32(hipStream_t, float*, const float*, const float*, int, int, int, bool, bool, int, const float*, const float*, float, bool);
hipError_t precond_grouped_f32(hipStream_t, const void*, int, int);
hipError_t precond_apply_grouped_f32(hipStream_t, const void*, int, int, float*, float*, long, float*, float*, float, float);
hipError_t precond_apply_only_f32(hipStream_t, const void*, int, int, float*, float*, float, float);
template <typename T>
hipError_t kl_clip_accum_t(hipStream_t, float*, const T*, const T*, long);
hipError_t triu_pack_f32(hipStream_t, float*, const float*, int);
hipError_t syevj_small_f32(hipStream_t, const float*, float*, float*, int, int, int, float);
hipError_t triu_unpack_f32(hipStream_t, float*, const float*, int);
hipError_t chol_diag_inv_f32(hipStream_t, float*, float*, int, int, int, int, float);
hipError_t gather_cols_f32(hipStream_t, const float*, const long*, const long*, float*, int, int, int, int);
hipError_t bucket_unpack_f32(hipStream_t, const float*, const void*, int, long);
hipError_t scatter_cols_f32(hipStream_t, float*, const long*, const long*, const float*, int, int, int, int);

}  // namespace kfac

namespace {

#define CHECK_OK(expr)                                                       \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));     \
  } while (0)

hipStream_t current_stream(const torch::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.device().index()).stream();
}

void check_gpu_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_out_factor(const torch::Tensor& out, int n) {
  check_gpu_contig(out, "out");
  TORCH_CHECK(out.scalar_type() == torch::kFloat32, "out must be fp32");
  TORCH_CHECK(
      out.dim() == 2 && out.size(0) == n && out.size(1) == n,
      "out must be (",
      n,
      ", ",
      n,
      "), got ",
      out.sizes());
}

template <typename F32Fn, typename Bf16Fn, typename F16Fn>
void dispatch_dtype(
    torch::ScalarType st,
    F32Fn f32fn,
    Bf16Fn bf16fn,
    F16Fn f16fn) {
  switch (st) {
    case torch::kFloat32:
      f32fn();
      break;
    case torch::kBFloat16:
      bf16fn();
      break;
    case torch::kFloat16:
      f16fn();
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype for kfac op: ", st);
  }
}

void cov_linear(
    torch::Tensor a,
    torch::Tensor out,
    bool bias,
    double beta,
    double coeff) {
  check_gpu_contig(a, "a");
  TORCH_CHECK(a.dim() == 2, "a must be 2D");
  int M = (int)a.size(0);
  int K = (int)a.size(1);
  int n = K + (bias ? 1 : 0);
  check_out_factor(out, n);
  auto stream = current_stream(a);
  float* outp = out.data_ptr<float>();
  dispatch_dtype(
      a.scalar_type(),
      [&] {
        CHECK_OK(kfac::cov_linear_t<float>(
            stream, a.data_ptr<float>(), K, M, K, bias, outp, (float)beta,
            (float)coeff));
      },
      [&] {
        // bf16 route: transpose once to [K][Mpad] so SYRK staging is
        // m-contiguous (16 B LDS writes) instead of column-fast scalar
        // scatter; the extra HBM round trip is trivial vs the n^2 MFMA
        // work (same argument as materialized im2col).
        const int Mpad = (M + 63) / 64 * 64;
        auto ws = torch::empty({(long)K, (long)Mpad}, a.options());
        CHECK_OK(kfac::transpose_to_bf16_t<__hip_bfloat16>(
            stream, (const __hip_bfloat16*)a.data_ptr(), K, M, K,
            (__hip_bfloat16*)ws.data_ptr(), Mpad));
        CHECK_OK(kfac::cov_trans_t(
            stream, (const __hip_bfloat16*)ws.data_ptr(), Mpad, M, K, bias,
            outp, (float)beta, (float)coeff));
      },
      [&] {
        CHECK_OK(kfac::cov_linear_t<__half>(
            stream, (const __half*)a.data_ptr(), K, M, K, bias, outp,
            (float)beta, (float)coeff));
      });
}

void cov_conv_a(
    torch::Tensor x,
    torch::Tensor out,
    int64_t kh,
    int64_t kw,
    int64_t sh,
    int64_t sw,
    int64_t ph,
    int64_t pw,
    bool bias,
    double beta,
    double coeff_scale) {
  check_gpu_contig(x, "x");
  TORCH_CHECK(x.dim() == 4, "x must be NCHW");
  int Nb = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      W = (int)x.size(3);
  int OH = (int)((H + 2 * ph - kh) / sh + 1);
  int OW = (int)((W + 2 * pw - kw) / sw + 1);
  TORCH_CHECK(OH > 0 && OW > 0, "empty conv output");
  int K = (int)(C * kh * kw);
  int n = K + (bias ? 1 : 0);
  check_out_factor(out, n);
  long m = (long)Nb * OH * OW;
  long s = (long)OH * OW;
  // patches/spatial (ones included) then cov scale 1/m:
  // coeff = coeff_scale / (m * s^2)   (see ops/reference.py cov_conv_a)
  double coeff = coeff_scale / ((double)m * (double)s * (double)s);
  auto stream = current_stream(x);
  float* outp = out.data_ptr<float>();
  // Materialize the patch matrix once (gather VALU cost paid once), then
  // run the flat-matrix SYRK with wide coalesced staging. The scratch
  // lives in the caching allocator; bytes are trivial vs 288 GB HBM3E.
  dispatch_dtype(
      x.scalar_type(),
      [&] {
        const int K_pad = (K + 7) / 8 * 8;
        auto scratch = torch::empty({m, (long)K_pad}, x.options());
        CHECK_OK(kfac::im2col_t<float>(
            stream, x.data_ptr<float>(), scratch.data_ptr<float>(), Nb, C, H,
            W, (int)kh, (int)kw, (int)sh, (int)sw, (int)ph, (int)pw, K_pad));
        CHECK_OK(kfac::cov_linear_t<float>(
            stream, scratch.data_ptr<float>(), K_pad, (int)m, K, bias, outp,
            (float)beta, (float)coeff));
      },
      [&] {
        // bf16: materialize the patch matrix TRANSPOSED ([K][Mpad]) —
        // im2col touches every element exactly once either way, and the
        // transposed image makes every SYRK staging write a contiguous
        // 16 B LDS store (see TransAcc in syrk.hip).
        const long Mpad = (m + 63) / 64 * 64;
        auto scratch = torch::empty({(long)K, Mpad}, x.options());
        CHECK_OK(kfac::im2col_tr_t<__hip_bfloat16>(
            stream, (const __hip_bfloat16*)x.data_ptr(),
            (__hip_bfloat16*)scratch.data_ptr(), Nb, C, H, W, (int)kh,
            (int)kw, (int)sh, (int)sw, (int)ph, (int)pw, (int)Mpad));
        CHECK_OK(kfac::cov_trans_t(
            stream, (const __hip_bfloat16*)scratch.data_ptr(), (int)Mpad,
            (int)m, K, bias, outp, (float)beta, (float)coeff));
      },
      [&] {
        const int K_pad = (K + 7) / 8 * 8;
        auto scratch = torch::empty({m, (long)K_pad}, x.options());
        CHECK_OK(kfac::im2col_t<__half>(
            stream, (const __half*)x.data_ptr(),
            (__half*)scratch.data_ptr(), Nb, C, H, W, (int)kh, (int)kw,
            (int)sh, (int)sw, (int)ph, (int)pw, K_pad));
        CHECK_OK(kfac::cov_linear_t<__half>(
            stream, (const __half*)scratch.data_ptr(), K_pad, (int)m, K,
            bias, outp, (float)beta, (float)coeff));
      });
}

void cov_conv_g(
    torch::Tensor g,
    torch::Tensor out,
    double beta,
    double coeff_scale) {
  check_gpu_contig(g, "g");
  TORCH_CHECK(g.dim() == 4, "g must be NCHW");
  int Nb = (int)g.size(0), C = (int)g.size(1), OH = (int)g.size(2),
      OW = (int)g.size(3);
  check_out_factor(out, C);
  long m = (long)Nb * OH * OW;
  long s = (long)OH * OW;
  double coeff = coeff_scale / ((double)m * (double)s * (double)s);
  auto stream = current_stream(g);
  float* outp = out.data_ptr<float>();
  dispatch_dtype(
      g.scalar_type(),
      [&] {
        CHECK_OK(kfac::cov_conv_g_t<float>(
            stream, g.data_ptr<float>(), Nb, C, OH, OW, outp, (float)beta,
            (float)coeff));
      },
      [&] {
        CHECK_OK(kfac::cov_conv_g_t<__hip_bfloat16>(
            stream, (const __hip_bfloat16*)g.data_ptr(), Nb, C, OH, OW, outp,
            (float)beta, (float)coeff));
      },
      [&] {
        CHECK_OK(kfac::cov_conv_g_t<__half>(
            stream, (const __half*)g.data_ptr(), Nb, C, OH, OW, outp,
            (float)beta, (float)coeff));
      });
}

// Shared tail of the eigen precondition chain:
//   given v2 (already divided/multiplied), out = (QG @ v2) @ QA^T.
// Association matches the reference (eigen.py:385: qg @ v2 @ qa.t() is
// left-associative) so numerics line up with the fp32 torch reference.
torch::Tensor eigen_tail(
    const torch::Tensor& v2,
    const torch::Tensor& qa,
    const torch::Tensor& qg,
    hipStream_t stream) {
  int m = (int)v2.size(0);
  int n = (int)v2.size(1);
  auto t2 = torch::empty_like(v2);
  // t2 = QG @ v2 : [m,m] x [m,n]
  CHECK_OK(kfac::gemm_f32(
      stream, t2.data_ptr<float>(), qg.data_ptr<float>(),
      v2.data_ptr<float>(), m, n, m, false, false, 0, nullptr, nullptr, 0.f,
      true));
  auto out = torch::empty_like(v2);
  // out = t2 @ QA^T : [m,n] x [n,n]^T
  CHECK_OK(kfac::gemm_f32(
      stream, out.data_ptr<float>(), t2.data_ptr<float>(),
      qa.data_ptr<float>(), m, n, n, false, true, 0, nullptr, nullptr, 0.f,
      true));
  return out;
}

torch::Tensor precond_eigen_fused(
    torch::Tensor grad,
    torch::Tensor qa,
    torch::Tensor qg,
    torch::Tensor dgda) {
  check_gpu_contig(grad, "grad");
  check_gpu_contig(qa, "qa");
  check_gpu_contig(qg, "qg");
  check_gpu_contig(dgda, "dgda");
  auto dtype = grad.scalar_type();
  auto g32 = grad.to(torch::kFloat32);
  int m = (int)g32.size(0);
  int n = (int)g32.size(1);
  TORCH_CHECK(qa.size(0) == n && qg.size(0) == m, "shape mismatch");
  auto stream = current_stream(grad);
  auto t1 = torch::empty_like(g32);
  // t1 = QG^T @ grad (reference association: (qg.t() @ grad) @ qa)
  CHECK_OK(kfac::gemm_f32(
      stream, t1.data_ptr<float>(), qg.data_ptr<float>(),
      g32.data_ptr<float>(), m, n, m, true, false, 0, nullptr, nullptr, 0.f,
      true));
  auto v2 = torch::empty_like(g32);
  // v2 = (t1 @ QA) * dgda   (epilogue-fused elementwise)
  CHECK_OK(kfac::gemm_f32(
      stream, v2.data_ptr<float>(), t1.data_ptr<float>(),
      qa.data_ptr<float>(), m, n, n, false, false, 1, dgda.data_ptr<float>(),
      nullptr, 0.f, true));
  return eigen_tail(v2, qa, qg, stream).to(dtype);
}

torch::Tensor precond_eigen(
    torch::Tensor grad,
    torch::Tensor qa,
    torch::Tensor qg,
    torch::Tensor dg,
    torch::Tensor da,
    double damping) {
  check_gpu_contig(grad, "grad");
  check_gpu_contig(qa, "qa");
  check_gpu_contig(qg, "qg");
  check_gpu_contig(dg, "dg");
  check_gpu_contig(da, "da");
  auto dtype = grad.scalar_type();
  auto g32 = grad.to(torch::kFloat32);
  int m = (int)g32.size(0);
  int n = (int)g32.size(1);
  auto stream = current_stream(grad);
  auto t1 = torch::empty_like(g32);
  // t1 = QG^T @ grad
  CHECK_OK(kfac::gemm_f32(
      stream, t1.data_ptr<float>(), qg.data_ptr<float>(),
      g32.data_ptr<float>(), m, n, m, true, false, 0, nullptr, nullptr, 0.f,
      true));
  auto v2 = torch::empty_like(g32);
  // v2 = (t1 @ QA) / (outer(dg, da) + damping)
  CHECK_OK(kfac::gemm_f32(
      stream, v2.data_ptr<float>(), t1.data_ptr<float>(),
      qa.data_ptr<float>(), m, n, n, false, false, 2, dg.data_ptr<float>(),
      da.data_ptr<float>(), (float)damping, true));
  return eigen_tail(v2, qa, qg, stream).to(dtype);
}

// Mirrors kfac::PrecondDesc in gemm.hip (10 x 8-byte fields).
struct PrecondDescHost {
  int64_t m, n;
  const float* grad;
  const float* qa;
  const float* qg;
  const float* dgda;
  float* s1;
  float* s2;
  float* out;
  int64_t tile_off;
  float* wgrad;
  float* bgrad;
  int64_t spad;
};
static_assert(sizeof(PrecondDescHost) == 104, "descriptor layout");

std::vector<torch::Tensor> precond_eigen_grouped(
    std::vector<torch::Tensor> grads,
    std::vector<torch::Tensor> qas,
    std::vector<torch::Tensor> qgs,
    std::vector<torch::Tensor> dgdas) {
  const int L = (int)grads.size();
  TORCH_CHECK(L > 0, "empty layer list");
  TORCH_CHECK(
      (int)qas.size() == L && (int)qgs.size() == L && (int)dgdas.size() == L,
      "list length mismatch");
  auto dev_opts =
      torch::TensorOptions().device(grads[0].device()).dtype(torch::kFloat32);

  std::vector<torch::Tensor> g32(L);
  std::vector<torch::ScalarType> dtypes(L);
  int64_t total = 0;
  int64_t tiles = 0;
  std::vector<int64_t> offsets(L);
  std::vector<int64_t> tile_offs(L);
  for (int l = 0; l < L; ++l) {
    check_gpu_contig(qas[l], "qa");
    check_gpu_contig(qgs[l], "qg");
    check_gpu_contig(dgdas[l], "dgda");
    TORCH_CHECK(grads[l].dim() == 2, "grad must be 2D");
    dtypes[l] = grads[l].scalar_type();
    g32[l] = grads[l].to(torch::kFloat32).contiguous();
    int64_t m = g32[l].size(0);
    int64_t n = g32[l].size(1);
    TORCH_CHECK(qgs[l].size(0) == m && qas[l].size(0) == n, "shape mismatch");
    offsets[l] = total;
    tile_offs[l] = tiles;
    total += m * n;
    // must match GBT=128 in gemm.hip
    tiles += (int64_t)((m + 127) / 128) * ((n + 127) / 128);
  }
  auto s1 = torch::empty({total}, dev_opts);
  auto s2 = torch::empty({total}, dev_opts);
  auto outbuf = torch::empty({total}, dev_opts);

  auto desc_cpu = torch::empty(
      {L * (int64_t)(sizeof(PrecondDescHost) / 8)},
      torch::TensorOptions().dtype(torch::kInt64).pinned_memory(true));
  auto* d = (PrecondDescHost*)desc_cpu.data_ptr<int64_t>();
  for (int l = 0; l < L; ++l) {
    d[l].m = g32[l].size(0);
    d[l].n = g32[l].size(1);
    d[l].grad = g32[l].data_ptr<float>();
    d[l].qa = qas[l].data_ptr<float>();
    d[l].qg = qgs[l].data_ptr<float>();
    d[l].dgda = dgdas[l].data_ptr<float>();
    d[l].s1 = s1.data_ptr<float>() + offsets[l];
    d[l].s2 = s2.data_ptr<float>() + offsets[l];
    d[l].out = outbuf.data_ptr<float>() + offsets[l];
    d[l].tile_off = tile_offs[l];
    d[l].wgrad = nullptr;
    d[l].bgrad = nullptr;
    d[l].spad = g32[l].size(1);
  }
  auto desc_dev = desc_cpu.to(grads[0].device(), /*non_blocking=*/true);
  auto stream = current_stream(grads[0]);
  CHECK_OK(kfac::precond_grouped_f32(
      stream, desc_dev.data_ptr<int64_t>(), L, (int)tiles));

  std::vector<torch::Tensor> outs(L);
  for (int l = 0; l < L; ++l) {
    auto view = outbuf
                    .narrow(0, offsets[l], d[l].m * d[l].n)
                    .view({d[l].m, d[l].n});
    outs[l] = (dtypes[l] == torch::kFloat32) ? view : view.to(dtypes[l]);
  }
  return outs;
}

// Fully-fused COMM-OPT precondition: gather module grads -> 4 GEMM
// stages -> device kl-clip -> scale+scatter back IN PLACE. ~9 kernel
// launches for the whole model (vs ~330 torch launches on the
// per-layer path). Returns the applied scale (1-elem tensor).
torch::Tensor precond_apply_grouped(
    std::vector<torch::Tensor> wgrads,
    std::vector<torch::Tensor> bgrads,  // 0-numel tensor = no bias
    std::vector<torch::Tensor> qas,
    std::vector<torch::Tensor> qgs,
    std::vector<torch::Tensor> dgdas,
    double kl_clip,
    double lr,
    // optional PRE-ACCUMULATED kl-clip dot contributions from layers
    // handled outside this call (e.g. the large layers routed through
    // the hipBLASLt xf32 chain); the scale covers both sets.
    c10::optional<torch::Tensor> accum_init) {
  const int L = (int)wgrads.size();
  TORCH_CHECK(L > 0, "empty layer list");
  TORCH_CHECK(
      (int)bgrads.size() == L && (int)qas.size() == L &&
          (int)qgs.size() == L && (int)dgdas.size() == L,
      "list length mismatch");
  auto dev_opts =
      torch::TensorOptions().device(wgrads[0].device()).dtype(torch::kFloat32);

  int64_t total = 0;
  int64_t tiles = 0;
  std::vector<int64_t> offsets(L);
  std::vector<int64_t> tile_offs(L);
  std::vector<int64_t> ms(L), ns(L);
  for (int l = 0; l < L; ++l) {
    check_gpu_contig(wgrads[l], "wgrad");
    check_gpu_contig(qas[l], "qa");
    check_gpu_contig(qgs[l], "qg");
    check_gpu_contig(dgdas[l], "dgda");
    TORCH_CHECK(
        wgrads[l].scalar_type() == torch::kFloat32,
        "fused path requires fp32 grads");
    const int64_t m = wgrads[l].size(0);
    const bool has_bias = bgrads[l].numel() > 0;
    if (has_bias) {
      check_gpu_contig(bgrads[l], "bgrad");
      TORCH_CHECK(bgrads[l].numel() == m, "bias grad size mismatch");
    }
    const int64_t n = wgrads[l].numel() / m + (has_bias ? 1 : 0);
    TORCH_CHECK(qgs[l].size(0) == m && qas[l].size(0) == n, "shape mismatch");
    TORCH_CHECK(
        dgdas[l].size(0) == m && dgdas[l].size(1) == n, "dgda shape");
    ms[l] = m;
    ns[l] = n;
    offsets[l] = total;
    tile_offs[l] = tiles;
    total += m * ((n + 3) / 4 * 4);
    tiles += (int64_t)((m + 127) / 128) * ((n + 127) / 128);
  }
  // grad/out buffers are zero-initialized: the row padding (spad > n)
  // participates in the flat kl-clip dot product and must contribute 0.
  auto gbuf = torch::zeros({total}, dev_opts);
  auto s1 = torch::empty({total}, dev_opts);
  auto s2 = torch::empty({total}, dev_opts);
  auto outbuf = torch::zeros({total}, dev_opts);
  auto work = torch::zeros({2}, dev_opts);  // [accum, scale]
  if (accum_init.has_value()) {
    check_gpu_contig(*accum_init, "accum_init");
    TORCH_CHECK(
        accum_init->scalar_type() == torch::kFloat32 &&
            accum_init->numel() == 1,
        "accum_init must be a 1-element fp32 tensor");
    work.narrow(0, 0, 1).copy_(accum_init->reshape({1}));
  }

  auto desc_cpu = torch::empty(
      {L * (int64_t)(sizeof(PrecondDescHost) / 8)},
      torch::TensorOptions().dtype(torch::kInt64).pinned_memory(true));
  auto* d = (PrecondDescHost*)desc_cpu.data_ptr<int64_t>();
  for (int l = 0; l < L; ++l) {
    d[l].m = ms[l];
    d[l].n = ns[l];
    d[l].grad = gbuf.data_ptr<float>() + offsets[l];
    d[l].qa = qas[l].data_ptr<float>();
    d[l].qg = qgs[l].data_ptr<float>();
    d[l].dgda = dgdas[l].data_ptr<float>();
    d[l].s1 = s1.data_ptr<float>() + offsets[l];
    d[l].s2 = s2.data_ptr<float>() + offsets[l];
    d[l].out = outbuf.data_ptr<float>() + offsets[l];
    d[l].tile_off = tile_offs[l];
    d[l].wgrad = wgrads[l].data_ptr<float>();
    d[l].bgrad =
        bgrads[l].numel() > 0 ? bgrads[l].data_ptr<float>() : nullptr;
    d[l].spad = (ns[l] + 3) / 4 * 4;
  }
  auto desc_dev = desc_cpu.to(wgrads[0].device(), /*non_blocking=*/true);
  auto stream = current_stream(wgrads[0]);
  CHECK_OK(kfac::precond_apply_grouped_f32(
      stream,
      desc_dev.data_ptr<int64_t>(),
      L,
      (int)tiles,
      gbuf.data_ptr<float>(),
      outbuf.data_ptr<float>(),
      total,
      work.data_ptr<float>(),
      work.data_ptr<float>() + 1,
      (float)kl_clip,
      (float)lr));
  return work.narrow(0, 1, 1);
}

// Fused allreduce-bucket unpack (K13): one kernel scatters the flat
// buffer into every member tensor (fp32 factors only; mixed dtypes
// keep the per-tensor copy path).
struct BucketDescHost {
  float* dst;
  long offset;
  long numel;
};
static_assert(sizeof(BucketDescHost) == 24, "bucket desc layout");

void bucket_unpack(torch::Tensor flat, std::vector<torch::Tensor> dsts) {
  check_gpu_contig(flat, "flat");
  TORCH_CHECK(flat.scalar_type() == torch::kFloat32, "fp32 only");
  const int n = (int)dsts.size();
  TORCH_CHECK(n > 0, "empty tensor list");
  auto desc_cpu = torch::empty(
      {n * 3},
      torch::TensorOptions().dtype(torch::kInt64).pinned_memory(true));
  auto* d = (BucketDescHost*)desc_cpu.data_ptr<int64_t>();
  long off = 0;
  for (int i = 0; i < n; ++i) {
    check_gpu_contig(dsts[i], "dst");
    TORCH_CHECK(dsts[i].scalar_type() == torch::kFloat32, "fp32 only");
    d[i].dst = dsts[i].data_ptr<float>();
    d[i].offset = off;
    d[i].numel = dsts[i].numel();
    off += d[i].numel;
  }
  TORCH_CHECK(off == flat.numel(), "bucket size mismatch");
  auto desc_dev = desc_cpu.to(flat.device(), /*non_blocking=*/true);
  CHECK_OK(kfac::bucket_unpack_f32(
      current_stream(flat), flat.data_ptr<float>(),
      desc_dev.data_ptr<int64_t>(), n, off));
}

// Batched column gather/scatter for the warm block-Jacobi rounds:
// out[p] = t[mat[p]][:, idx[p]] without materializing index grids.
torch::Tensor gather_cols(
    torch::Tensor t,     // (B, rows, n) fp32 contiguous
    torch::Tensor mat,   // (p,) int64
    torch::Tensor idx) {  // (p, m) int64
  check_gpu_contig(t, "t");
  check_gpu_contig(mat, "mat");
  check_gpu_contig(idx, "idx");
  TORCH_CHECK(t.dim() == 3 && t.scalar_type() == torch::kFloat32, "t");
  TORCH_CHECK(
      mat.scalar_type() == torch::kInt64 &&
          idx.scalar_type() == torch::kInt64,
      "int64 indices required");
  const int p = (int)mat.size(0);
  const int rows = (int)t.size(1);
  const int n = (int)t.size(2);
  const int m = (int)idx.size(1);
  auto out = torch::empty(
      {(long)p, (long)rows, (long)m},
      torch::TensorOptions().device(t.device()).dtype(torch::kFloat32));
  CHECK_OK(kfac::gather_cols_f32(
      current_stream(t), t.data_ptr<float>(), mat.data_ptr<long>(),
      idx.data_ptr<long>(), out.data_ptr<float>(), p, rows, n, m));
  return out;
}

void scatter_cols(
    torch::Tensor t,
    torch::Tensor mat,
    torch::Tensor idx,
    torch::Tensor src) {
  check_gpu_contig(t, "t");
  check_gpu_contig(mat, "mat");
  check_gpu_contig(idx, "idx");
  check_gpu_contig(src, "src");
  TORCH_CHECK(
      t.dim() == 3 && t.scalar_type() == torch::kFloat32 &&
          src.scalar_type() == torch::kFloat32,
      "t/src must be 3D fp32");
  TORCH_CHECK(
      mat.scalar_type() == torch::kInt64 &&
          idx.scalar_type() == torch::kInt64,
      "int64 indices required");
  const int p = (int)mat.size(0);
  const int rows = (int)t.size(1);
  const int n = (int)t.size(2);
  const int m = (int)idx.size(1);
  TORCH_CHECK(
      src.dim() == 3 && src.size(0) == p && src.size(1) == rows &&
          src.size(2) == m,
      "src shape");
  CHECK_OK(kfac::scatter_cols_f32(
      current_stream(t), t.data_ptr<float>(), mat.data_ptr<long>(),
      idx.data_ptr<long>(), src.data_ptr<float>(), p, rows, n, m));
}

// One blocked-Cholesky diagonal step: factor the m x m block of each
// matrix at (j, j) in LDS (lower, diag floored at eps) and write its
// triangular inverse into dinv (B, 128, 128).  The O(n^3) panel /
// trailing GEMMs around it are driven from Python (ops/blocked.py).
void chol_diag_inv(
    torch::Tensor a,
    torch::Tensor dinv,
    int64_t j,
    int64_t m,
    double eps) {
  check_gpu_contig(a, "a");
  check_gpu_contig(dinv, "dinv");
  TORCH_CHECK(
      a.dim() == 3 && a.size(1) == a.size(2), "a must be (B, n, n)");
  TORCH_CHECK(a.scalar_type() == torch::kFloat32, "fp32 only");
  const int B = (int)a.size(0);
  const int n = (int)a.size(1);
  TORCH_CHECK(m >= 1 && m <= 128, "block size must be in [1, 128]");
  TORCH_CHECK(j >= 0 && j + m <= n, "block out of range");
  TORCH_CHECK(
      dinv.dim() == 3 && dinv.size(0) == B && dinv.size(1) == 128 &&
          dinv.size(2) == 128,
      "dinv must be (B, 128, 128)");
  CHECK_OK(kfac::chol_diag_inv_f32(
      current_stream(a), a.data_ptr<float>(), dinv.data_ptr<float>(), B, n,
      (int)j, (int)m, (float)eps));
}

// Raw MFMA GEMM: C = op(A) @ op(B), fp32 in/out.  split=true runs the
// bf16x3 split-precision path (hi/lo bf16 decomposition, fp32-class
// accuracy at bf16 MFMA rates).  Building block for the QDWH polar
// iterations and the batched Cholesky host loops.
torch::Tensor gemm(
    torch::Tensor a,
    torch::Tensor b,
    bool ta,
    bool tb,
    bool split) {
  check_gpu_contig(a, "a");
  check_gpu_contig(b, "b");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm expects 2D");
  TORCH_CHECK(
      a.scalar_type() == torch::kFloat32 &&
          b.scalar_type() == torch::kFloat32,
      "gemm: fp32 only");
  const int M = (int)(ta ? a.size(1) : a.size(0));
  const int K = (int)(ta ? a.size(0) : a.size(1));
  const int Kb = (int)(tb ? b.size(1) : b.size(0));
  const int N = (int)(tb ? b.size(0) : b.size(1));
  TORCH_CHECK(K == Kb, "gemm: inner dims mismatch");
  auto out = torch::empty(
      {(long)M, (long)N},
      torch::TensorOptions().device(a.device()).dtype(torch::kFloat32));
  CHECK_OK(kfac::gemm_f32(
      current_stream(a), out.data_ptr<float>(), a.data_ptr<float>(),
      b.data_ptr<float>(), M, N, K, ta, tb, 0, nullptr, nullptr, 0.f,
      split));
  return out;
}

// Fused kl-clip + scaled in-place grad write for ALREADY-preconditioned
// per-layer gradients (the HYBRID/MEM-OPT path: grad workers ran the
// grouped chain, receivers got broadcasts; clip+apply remain).  accum
// must be pre-filled with any per-layer contributions from layers not
// in this call.  3 launches replace ~3 per layer.
torch::Tensor apply_scaled_grouped(
    std::vector<torch::Tensor> precons,
    std::vector<torch::Tensor> wgrads,
    std::vector<torch::Tensor> bgrads,
    torch::Tensor accum,
    double kl_clip,
    double lr) {
  const int L = (int)precons.size();
  TORCH_CHECK(L > 0, "empty layer list");
  TORCH_CHECK(
      (int)wgrads.size() == L && (int)bgrads.size() == L, "length mismatch");
  check_gpu_contig(accum, "accum");
  TORCH_CHECK(accum.scalar_type() == torch::kFloat32, "accum fp32");
  int64_t tiles = 0;
  std::vector<int64_t> tile_offs(L);
  for (int l = 0; l < L; ++l) {
    check_gpu_contig(precons[l], "precon");
    check_gpu_contig(wgrads[l], "wgrad");
    TORCH_CHECK(
        precons[l].scalar_type() == torch::kFloat32 &&
            wgrads[l].scalar_type() == torch::kFloat32,
        "fp32 required");
    const int64_t m = precons[l].size(0);
    const int64_t n = precons[l].size(1);
    const bool has_bias = bgrads[l].numel() > 0;
    TORCH_CHECK(
        wgrads[l].numel() == m * (n - (has_bias ? 1 : 0)),
        "wgrad size mismatch");
    if (has_bias) {
      check_gpu_contig(bgrads[l], "bgrad");
      TORCH_CHECK(bgrads[l].numel() == m, "bias size mismatch");
    }
    tile_offs[l] = tiles;
    tiles += (int64_t)((m + 127) / 128) * ((n + 127) / 128);
  }
  auto desc_cpu = torch::empty(
      {L * (int64_t)(sizeof(PrecondDescHost) / 8)},
      torch::TensorOptions().dtype(torch::kInt64).pinned_memory(true));
  auto* d = (PrecondDescHost*)desc_cpu.data_ptr<int64_t>();
  for (int l = 0; l < L; ++l) {
    d[l].m = precons[l].size(0);
    d[l].n = precons[l].size(1);
    d[l].grad = nullptr;
    d[l].qa = nullptr;
    d[l].qg = nullptr;
    d[l].dgda = nullptr;
    d[l].s1 = nullptr;
    d[l].s2 = nullptr;
    d[l].out = precons[l].data_ptr<float>();
    d[l].tile_off = tile_offs[l];
    d[l].wgrad = wgrads[l].data_ptr<float>();
    d[l].bgrad =
        bgrads[l].numel() > 0 ? bgrads[l].data_ptr<float>() : nullptr;
    d[l].spad = precons[l].size(1);
  }
  auto desc_dev = desc_cpu.to(precons[0].device(), /*non_blocking=*/true);
  auto scale = torch::empty(
      {1},
      torch::TensorOptions().device(accum.device()).dtype(torch::kFloat32));
  CHECK_OK(kfac::precond_apply_only_f32(
      current_stream(accum),
      desc_dev.data_ptr<int64_t>(),
      L,
      (int)tiles,
      accum.data_ptr<float>(),
      scale.data_ptr<float>(),
      (float)kl_clip,
      (float)lr));
  return scale;
}

torch::Tensor precond_inverse(
    torch::Tensor grad,
    torch::Tensor a_inv,
    torch::Tensor g_inv) {
  check_gpu_contig(grad, "grad");
  check_gpu_contig(a_inv, "a_inv");
  check_gpu_contig(g_inv, "g_inv");
  auto dtype = grad.scalar_type();
  auto g32 = grad.to(torch::kFloat32);
  auto ai = a_inv.scalar_type() == torch::kFloat32 ? a_inv
                                                   : a_inv.to(torch::kFloat32);
  auto gi = g_inv.scalar_type() == torch::kFloat32 ? g_inv
                                                   : g_inv.to(torch::kFloat32);
  int m = (int)g32.size(0);
  int n = (int)g32.size(1);
  auto stream = current_stream(grad);
  auto t1 = torch::empty_like(g32);
  // t1 = G^-1 @ grad
  CHECK_OK(kfac::gemm_f32(
      stream, t1.data_ptr<float>(), gi.data_ptr<float>(),
      g32.data_ptr<float>(), m, n, m, false, false, 0, nullptr, nullptr,
      0.f, false));
  auto out = torch::empty_like(g32);
  // out = t1 @ A^-1
  CHECK_OK(kfac::gemm_f32(
      stream, out.data_ptr<float>(), t1.data_ptr<float>(),
      ai.data_ptr<float>(), m, n, n, false, false, 0, nullptr, nullptr, 0.f,
      false));
  return out.to(dtype);
}

void kl_clip_accum(
    torch::Tensor accum,
    torch::Tensor precon,
    torch::Tensor grad) {
  check_gpu_contig(accum, "accum");
  check_gpu_contig(precon, "precon");
  check_gpu_contig(grad, "grad");
  TORCH_CHECK(accum.scalar_type() == torch::kFloat32, "accum must be fp32");
  TORCH_CHECK(precon.scalar_type() == grad.scalar_type(), "dtype mismatch");
  TORCH_CHECK(precon.numel() == grad.numel(), "numel mismatch");
  auto stream = current_stream(accum);
  long n = precon.numel();
  dispatch_dtype(
      precon.scalar_type(),
      [&] {
        CHECK_OK(kfac::kl_clip_accum_t<float>(
            stream, accum.data_ptr<float>(), precon.data_ptr<float>(),
            grad.data_ptr<float>(), n));
      },
      [&] {
        CHECK_OK(kfac::kl_clip_accum_t<__hip_bfloat16>(
            stream, accum.data_ptr<float>(),
            (const __hip_bfloat16*)precon.data_ptr(),
            (const __hip_bfloat16*)grad.data_ptr(), n));
      },
      [&] {
        CHECK_OK(kfac::kl_clip_accum_t<__half>(
            stream, accum.data_ptr<float>(), (const __half*)precon.data_ptr(),
            (const __half*)grad.data_ptr(), n));
      });
}

// Batched symmetric eigensolver via rocSOLVER syevd.
//
// NOTE (round 2): the round-1 version captured the ~50k-launch
// tridiagonalization sequence into a hipGraph and replayed it on later
// phases.  That is UNSOUND: syevd's tridiagonal eigeniteration issues a
// data-dependent number of kernels, so a captured sequence replayed on
// different data computes garbage — observed as rec errors ~1e2 on
// small G-factor groups whose spectra needed more iterations than the
// capture-time data (profiles/jacobi_warm.md).  The capture machinery
// is deleted; the warm-started block-Jacobi path (ops/warm_eigh.py)
// supersedes its performance role, and this direct call serves cold
// starts and fallbacks.
struct SyevdEntry {
  torch::Tensor a;     // persistent input/output (B, n, n)
  torch::Tensor w;     // eigenvalues (B, n)
  torch::Tensor e;     // tridiagonal workspace (B, n)
  torch::Tensor info;  // (B,) int32
};

rocblas_status run_syevd(
    rocblas_handle handle,
    SyevdEntry& ent,
    int n,
    int B) {
  return rocsolver_ssyevd_strided_batched(
      handle,
      rocblas_evect_original,
      rocblas_fill_upper,
      n,
      ent.a.data_ptr<float>(),
      n,
      (rocblas_stride)n * n,
      ent.w.data_ptr<float>(),
      (rocblas_stride)n,
      ent.e.data_ptr<float>(),
      (rocblas_stride)n,
      ent.info.data_ptr<int>(),
      B);
}

std::tuple<torch::Tensor, torch::Tensor> syevd_batched(torch::Tensor stack) {
  check_gpu_contig(stack, "stack");
  TORCH_CHECK(
      stack.dim() == 3 && stack.size(1) == stack.size(2),
      "stack must be (B, n, n)");
  TORCH_CHECK(stack.scalar_type() == torch::kFloat32, "fp32 only");
  const int B = (int)stack.size(0);
  const int n = (int)stack.size(1);

  static std::mutex mu;
  static rocblas_handle handle = nullptr;
  static std::map<std::tuple<int, int, int>, SyevdEntry> cache;
  std::lock_guard<std::mutex> lock(mu);
  if (handle == nullptr) {
    TORCH_CHECK(
        rocblas_create_handle(&handle) == rocblas_status_success,
        "rocblas_create_handle failed");
  }
  auto stream = current_stream(stack);

  const auto key = std::make_tuple((int)stack.device().index(), B, n);
  auto it = cache.find(key);
  if (it == cache.end()) {
    SyevdEntry ent;
    ent.a = torch::empty_like(stack);
    ent.w = torch::empty({B, (long)n}, stack.options());
    ent.e = torch::empty({B, (long)n}, stack.options());
    ent.info = torch::empty(
        {B},
        torch::TensorOptions().device(stack.device()).dtype(torch::kInt32));
    it = cache.emplace(key, std::move(ent)).first;
  }
  SyevdEntry& ent = it->second;
  ent.a.copy_(stack);
  rocblas_set_stream(handle, stream);
  auto st = run_syevd(handle, ent, n, B);
  TORCH_CHECK(st == rocblas_status_success, "rocsolver syevd failed: ", st);
  // syevd's tridiagonal QL iteration can FAIL to converge on hard
  // spectra (observed on fully-decayed scalar-identity G factors with
  // denormal off-diagonals): info != 0 and the outputs are garbage.
  // Re-solve failed matrices with the (slower, robust) Jacobi solver.
  auto info_host = ent.info.cpu();
  bool any_failed = false;
  for (int i = 0; i < B; ++i) {
    if (info_host[i].item<int>() != 0) {
      any_failed = true;
      break;
    }
  }
  if (any_failed) {
    auto a2 = stack.clone();
    auto residual = torch::empty({B}, stack.options());
    auto opts_i =
        torch::TensorOptions().device(stack.device()).dtype(torch::kInt32);
    auto n_sweeps = torch::empty({B}, opts_i);
    auto status = rocsolver_ssyevj_strided_batched(
        handle,
        rocblas_esort_ascending,
        rocblas_evect_original,
        rocblas_fill_upper,
        n,
        a2.data_ptr<float>(),
        n,
        (rocblas_stride)n * n,
        0.0f,
        residual.data_ptr<float>(),
        (rocblas_int)100,
        n_sweeps.data_ptr<int>(),
        ent.w.data_ptr<float>(),
        (rocblas_stride)n,
        ent.info.data_ptr<int>(),
        B);
    TORCH_CHECK(
        status == rocblas_status_success,
        "rocsolver syevj fallback failed: ",
        status);
    return {ent.w.clone(), a2};
  }
  // Column-major eigenvectors: the row-major clone holds V^T per matrix.
  return {ent.w.clone(), ent.a.clone()};
}

// Batched Jacobi symmetric eigensolver via rocSOLVER syevj: a few large
// batched kernels per sweep instead of syevd's ~50k tiny tridiagonalization
// launches — faster for K-FAC's many same-size factors and far friendlier
// to the async-inverse worker thread (launch-lock contention).
std::tuple<torch::Tensor, torch::Tensor> eigh_jacobi(
    torch::Tensor stack,
    double abstol,
    int64_t max_sweeps) {
  check_gpu_contig(stack, "stack");
  TORCH_CHECK(
      stack.dim() == 3 && stack.size(1) == stack.size(2),
      "stack must be (B, n, n)");
  TORCH_CHECK(stack.scalar_type() == torch::kFloat32, "fp32 only");
  const int B = (int)stack.size(0);
  const int n = (int)stack.size(1);

  // One mutex across the whole call: the handle's bound stream must not
  // change between set_stream and the launch if another thread (e.g.
  // the async-inverse worker) calls in concurrently.
  static std::mutex mu;
  static rocblas_handle handle = nullptr;
  std::lock_guard<std::mutex> lock(mu);
  if (handle == nullptr) {
    TORCH_CHECK(
        rocblas_create_handle(&handle) == rocblas_status_success,
        "rocblas_create_handle failed");
  }
  auto stream = current_stream(stack);
  rocblas_set_stream(handle, stream);

  // rocSOLVER overwrites A with the eigenvectors (column-major -> the
  // row-major view holds V^T).
  auto a = stack.clone();
  auto w = torch::empty({B, (long)n}, stack.options());
  auto residual = torch::empty({B}, stack.options());
  auto opts_i =
      torch::TensorOptions().device(stack.device()).dtype(torch::kInt32);
  auto n_sweeps = torch::empty({B}, opts_i);
  auto info = torch::empty({B}, opts_i);

  auto status = rocsolver_ssyevj_strided_batched(
      handle,
      rocblas_esort_ascending,
      rocblas_evect_original,
      rocblas_fill_upper,
      n,
      a.data_ptr<float>(),
      n,
      (rocblas_stride)n * n,
      (float)abstol,
      residual.data_ptr<float>(),
      (rocblas_int)max_sweeps,
      n_sweeps.data_ptr<int>(),
      w.data_ptr<float>(),
      (rocblas_stride)n,
      info.data_ptr<int>(),
      B);
  TORCH_CHECK(
      status == rocblas_status_success, "rocsolver syevj failed: ", status);
  // V^T (row-major view) -> V with eigenvectors as columns.
  auto q = a.transpose(1, 2).contiguous();
  return {w, q};
}

// Hand-written LDS-resident batched Jacobi eigensolver (n <= 128): one
// kernel launch per group; eigenvalues UNSORTED (K-FAC is order-
// invariant) and clamped >= 0.
std::tuple<torch::Tensor, torch::Tensor> syevj_small(
    torch::Tensor stack,
    int64_t max_sweeps,
    double tol) {
  check_gpu_contig(stack, "stack");
  TORCH_CHECK(
      stack.dim() == 3 && stack.size(1) == stack.size(2),
      "stack must be (B, n, n)");
  TORCH_CHECK(stack.scalar_type() == torch::kFloat32, "fp32 only");
  const int B = (int)stack.size(0);
  const int n = (int)stack.size(1);
  TORCH_CHECK(n <= 64, "syevj_small supports n <= 64");
  auto w = torch::empty({B, (long)n}, stack.options());
  auto v = torch::empty_like(stack);
  CHECK_OK(kfac::syevj_small_f32(
      current_stream(stack), stack.data_ptr<float>(), w.data_ptr<float>(),
      v.data_ptr<float>(), n, B, (int)max_sweeps, (float)tol));
  return {w, v};
}

torch::Tensor triu_pack(torch::Tensor x) {
  check_gpu_contig(x, "x");
  TORCH_CHECK(x.dim() == 2 && x.size(0) == x.size(1), "x must be square");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "triu_pack: fp32 only");
  int n = (int)x.size(0);
  auto out = torch::empty(
      {(long)n * (n + 1) / 2},
      x.options());
  CHECK_OK(kfac::triu_pack_f32(
      current_stream(x), out.data_ptr<float>(), x.data_ptr<float>(), n));
  return out;
}

torch::Tensor triu_unpack(torch::Tensor v, int64_t n) {
  check_gpu_contig(v, "v");
  TORCH_CHECK(v.scalar_type() == torch::kFloat32, "triu_unpack: fp32 only");
  TORCH_CHECK(v.numel() == n * (n + 1) / 2, "packed size mismatch");
  auto out = torch::empty({n, n}, v.options());
  CHECK_OK(kfac::triu_unpack_f32(
      current_stream(v), out.data_ptr<float>(), v.data_ptr<float>(), (int)n));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "kfac_amd CDNA4 (gfx950) kernels";
  m.def("cov_linear", &cov_linear, "fused [a,1]^T[a,1] SYRK accumulation");
  m.def("cov_conv_a", &cov_conv_a, "fused im2col+SYRK A-factor accumulation");
  m.def("cov_conv_g", &cov_conv_g, "fused NCHW-transpose SYRK G-factor");
  m.def("precond_eigen_fused", &precond_eigen_fused, "Kronecker precondition (prediv)");
  m.def("precond_eigen", &precond_eigen, "Kronecker precondition (dg/da)");
  m.def("precond_inverse", &precond_inverse, "G^-1 grad A^-1");
  m.def(
      "bucket_unpack",
      &bucket_unpack,
      "fused flat-bucket scatter into member tensors (K13)");
  m.def(
      "gather_cols",
      &gather_cols,
      "batched per-pair column gather (warm Jacobi rounds)");
  m.def(
      "scatter_cols",
      &scatter_cols,
      "batched per-pair column scatter (warm Jacobi rounds)");
  m.def(
      "chol_diag_inv",
      &chol_diag_inv,
      "in-LDS Cholesky + triangular inverse of one diagonal block",
      pybind11::arg("a"),
      pybind11::arg("dinv"),
      pybind11::arg("j"),
      pybind11::arg("m"),
      pybind11::arg("eps") = 1e-30);
  m.def(
      "gemm",
      &gemm,
      "C = op(A) op(B), fp32; split=true -> bf16x3 MFMA path",
      pybind11::arg("a"),
      pybind11::arg("b"),
      pybind11::arg("ta") = false,
      pybind11::arg("tb") = false,
      pybind11::arg("split") = true);
  m.def(
      "apply_scaled_grouped",
      &apply_scaled_grouped,
      "fused kl-clip + scaled in-place apply for preconditioned grads");
  m.def(
      "precond_apply_grouped",
      &precond_apply_grouped,
      "fused gather->precondition->kl-clip->scaled in-place grad update",
      pybind11::arg("wgrads"),
      pybind11::arg("bgrads"),
      pybind11::arg("qas"),
      pybind11::arg("qgs"),
      pybind11::arg("dgdas"),
      pybind11::arg("kl_clip"),
      pybind11::arg("lr"),
      pybind11::arg("accum_init") = pybind11::none());
  m.def(
      "precond_eigen_grouped",
      &precond_eigen_grouped,
      "whole precondition chain for all layers in 4 launches");
  m.def("kl_clip_accum", &kl_clip_accum, "device-side kl-clip accumulation");
  m.def(
      "syevj_small",
      &syevj_small,
      "LDS-resident batched Jacobi eigensolver (n <= 64, unsorted)",
      pybind11::arg("stack"),
      pybind11::arg("max_sweeps") = 20,
      pybind11::arg("tol") = 1e-5);
  m.def(
      "syevd_batched",
      &syevd_batched,
      "Batched symmetric eigendecomposition (rocSOLVER syevd, hipGraph "
      "replay); returns (w, Vt)");
  m.def(
      "eigh_jacobi",
      &eigh_jacobi,
      "batched Jacobi symmetric eigendecomposition (rocSOLVER syevj)",
      pybind11::arg("stack"),
      pybind11::arg("abstol") = 0.0,
      pybind11::arg("max_sweeps") = 100);
  m.def("triu_pack", &triu_pack, "pack upper triangle");
  m.def("triu_unpack", &triu_unpack, "unpack upper triangle");
}
