// Python bindings for the distar_amd HIP/CDNA4 kernels (_hip_ops).
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <c10/hip/HIPStream.h>

extern "C" __global__ void lambda_return_kernel(
    const float*, const float*, const float*, const float*, float*, int, int);
extern "C" __global__ void vtrace_kernel(
    const float*, const float*, const float*, const float*, const float*,
    const float*, float*, int, int);

using bf16_t = __hip_bfloat16;
extern "C" __global__ void lnlstm_forward_coop_kernel(
    const float*, const float*, const float*, const bf16_t*,
    const float*, const float*, const float*, const float*,
    float*, float*, float*, float*, bf16_t*, float*, float*, float*,
    int, int, int);
extern "C" __global__ void lnlstm_forward_kernel(
    const float*, const float*, const float*, const bf16_t*,
    const float*, const float*, const float*, const float*,
    float*, float*, float*, float*, int, int, int);
extern "C" __global__ void lnlstm_backward_kernel(
    const float*, const float*, const float*,
    const float*, const float*, const float*, const float*, const float*,
    const bf16_t*, const float*, const float*, const float*, const float*,
    float*, float*, float*, float*, float*, float*, float*, float*,
    int, int, int);

namespace {

inline void check_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on device");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

torch::Tensor lambda_return_scan(torch::Tensor rewards, torch::Tensor gammas,
                                 torch::Tensor values_tp1, torch::Tensor lambdas) {
  check_2d(rewards, "rewards");
  check_2d(gammas, "gammas");
  check_2d(values_tp1, "values_tp1");
  check_2d(lambdas, "lambdas");
  int64_t T = rewards.size(0), B = rewards.numel() / T;
  auto out = torch::empty_like(rewards);
  int threads = 256;
  int blocks = std::min<int64_t>((B + threads - 1) / threads, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lambda_return_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(),
                     rewards.data_ptr<float>(), gammas.data_ptr<float>(),
                     values_tp1.data_ptr<float>(), lambdas.data_ptr<float>(),
                     out.data_ptr<float>(), (int)T, (int)B);
  return out;
}

torch::Tensor vtrace_scan(torch::Tensor clipped_rhos, torch::Tensor clipped_cs,
                          torch::Tensor rewards, torch::Tensor values,
                          torch::Tensor gammas, torch::Tensor lambdas) {
  check_2d(clipped_rhos, "clipped_rhos");
  check_2d(rewards, "rewards");
  check_2d(values, "values");
  int64_t T = rewards.size(0), B = rewards.numel() / T;
  TORCH_CHECK(values.size(0) == T + 1, "values must be (T+1, B)");
  auto out = torch::empty_like(values);
  int threads = 256;
  int blocks = std::min<int64_t>((B + threads - 1) / threads, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(vtrace_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(),
                     clipped_rhos.data_ptr<float>(), clipped_cs.data_ptr<float>(),
                     rewards.data_ptr<float>(), values.data_ptr<float>(),
                     gammas.data_ptr<float>(), lambdas.data_ptr<float>(),
                     out.data_ptr<float>(), (int)T, (int)B);
  return out;
}

extern "C" __global__ void upsample2x_fwd_f32(const float*, float*, int, int, int);
extern "C" __global__ void upsample2x_fwd_bf16(const __hip_bfloat16*, __hip_bfloat16*, int, int, int);
extern "C" __global__ void upsample2x_fwd_bf16_fast(const __hip_bfloat16*, __hip_bfloat16*, int, int, int);
extern "C" __global__ void upsample2x_bwd_f32(const float*, float*, int, int, int);
extern "C" __global__ void upsample2x_bwd_bf16(const __hip_bfloat16*, __hip_bfloat16*, int, int, int);
extern "C" __global__ void upsample2x_bwd_bf16_fast(const __hip_bfloat16*, __hip_bfloat16*, int, int, int);
extern "C" __global__ void maxpool2x2_fwd_vec_kernel(
    const __hip_bfloat16*, __hip_bfloat16*, unsigned char*, long, int, int);
extern "C" __global__ void maxpool2x2_bwd_vec_kernel(
    const __hip_bfloat16*, const unsigned char*, __hip_bfloat16*,
    long, int, int);

torch::Tensor upsample2x(torch::Tensor input) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous() && input.dim() == 4);
  int64_t N = input.size(0), C = input.size(1), H = input.size(2), W = input.size(3);
  auto out = torch::empty({N, C, H * 2, W * 2}, input.options());
  long total = N * C * H * 2 * W * 2;
  int threads = 256;
  int blocks = std::min<long>((total + threads - 1) / threads, 8192);
  auto stream = c10::hip::getCurrentHIPStream();
  if (input.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(upsample2x_fwd_f32, dim3(blocks), dim3(threads), 0,
                       stream.stream(), input.data_ptr<float>(),
                       out.data_ptr<float>(), (int)(N * C), (int)H, (int)W);
  } else if (input.scalar_type() == torch::kBFloat16) {
    long groups = N * C * H * 2 * (W * 2 / 8);
    int gb = (int)std::min<long>((groups + threads - 1) / threads, 8192);
    hipLaunchKernelGGL(upsample2x_fwd_bf16_fast, dim3(gb), dim3(threads), 0,
                       stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(input.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       (int)(N * C), (int)H, (int)W);
  } else {
    TORCH_CHECK(false, "upsample2x: fp32/bf16 only");
  }
  return out;
}

torch::Tensor upsample2x_backward(torch::Tensor gout) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous() && gout.dim() == 4);
  int64_t N = gout.size(0), C = gout.size(1), H2 = gout.size(2), W2 = gout.size(3);
  int64_t H = H2 / 2, W = W2 / 2;
  auto gin = torch::empty({N, C, H, W}, gout.options());
  long total = N * C * H * W;
  int threads = 256;
  int blocks = std::min<long>((total + threads - 1) / threads, 8192);
  auto stream = c10::hip::getCurrentHIPStream();
  if (gout.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(upsample2x_bwd_f32, dim3(blocks), dim3(threads), 0,
                       stream.stream(), gout.data_ptr<float>(),
                       gin.data_ptr<float>(), (int)(N * C), (int)H, (int)W);
  } else if (gout.scalar_type() == torch::kBFloat16) {
    // fast path: constant 4x4 tap weights, generic per-px at borders
    long groups = N * C * H * (W / 4);
    int gblocks = (int)std::min<long>((groups + threads - 1) / threads, 8192);
    hipLaunchKernelGGL(upsample2x_bwd_bf16_fast, dim3(gblocks), dim3(threads),
                       0, stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(gout.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(gin.data_ptr()),
                       (int)(N * C), (int)H, (int)W);
  } else {
    TORCH_CHECK(false, "upsample2x_backward: fp32/bf16 only");
  }
  return gin;
}

constexpr int kNT = 512;   // must match NT in lnlstm.hip

std::vector<torch::Tensor> lnlstm_forward(
    torch::Tensor igates, torch::Tensor h0, torch::Tensor c0,
    torch::Tensor w_hh_bf16, torch::Tensor lnh_w, torch::Tensor lnh_b,
    torch::Tensor lnc_w, torch::Tensor lnc_b) {
  check_2d(igates, "igates");
  TORCH_CHECK(w_hh_bf16.scalar_type() == torch::kBFloat16, "w_hh must be bf16");
  TORCH_CHECK(w_hh_bf16.is_contiguous());
  int64_t T = igates.size(0), B = igates.size(1), G = igates.size(2);
  int64_t H = G / 4;
  TORCH_CHECK(H <= kNT && G <= 4 * kNT, "hidden size too large for kernel");
  auto opt = igates.options();
  auto h_all = torch::empty({T + 1, B, H}, opt);
  auto c_all = torch::empty({T + 1, B, H}, opt);
  auto hgates_raw = torch::empty({T, B, G}, opt);
  auto cellraw = torch::empty({T, B, H}, opt);
  size_t lds = sizeof(float) * ((H + 1) / 2 + G + H + kNT);
  auto stream = c10::hip::getCurrentHIPStream();
  static const bool use_coop = []() {
    const char* v = getenv("DISTAR_AMD_LSTM_COOP");
    return v && v[0] == '1';
  }();
  if (use_coop && B <= 64 && H % 32 == 0) {
    // small-batch cooperative variant: spreads each timestep's h @ W_hh^T
    // across the chip as an MFMA matmul.  MEASURED NULL RESULT (r2w):
    // grid.sync() costs ~50us on gfx950, and 3 syncs x 64 steps eat the
    // occupancy win — kept behind DISTAR_AMD_LSTM_COOP=1 as evidence.
    int64_t Mt = (B + 15) / 16, Nt = G / 16;
    int blocks = (int)((Mt * Nt + 3) / 4);
    auto bopt = igates.options().dtype(torch::kBFloat16);
    auto h_bf = torch::zeros({Mt * 16, H}, bopt);
    h_bf.narrow(0, 0, B).copy_(h0);
    auto o_ws = torch::empty({B, H}, igates.options());
    auto hstats = torch::zeros({2, B, 2}, igates.options());
    auto cstats = torch::zeros({2, B, 2}, igates.options());
    h_all[0].copy_(h0);
    c_all[0].copy_(c0);
    const float* igp = igates.data_ptr<float>();
    const float* h0p = h0.data_ptr<float>();
    const float* c0p = c0.data_ptr<float>();
    const __hip_bfloat16* wp2 =
        reinterpret_cast<const __hip_bfloat16*>(w_hh_bf16.data_ptr());
    const float* lhw = lnh_w.data_ptr<float>();
    const float* lhb = lnh_b.data_ptr<float>();
    const float* lcw = lnc_w.data_ptr<float>();
    const float* lcb = lnc_b.data_ptr<float>();
    float* hap = h_all.data_ptr<float>();
    float* cap = c_all.data_ptr<float>();
    float* hgp = hgates_raw.data_ptr<float>();
    float* crp = cellraw.data_ptr<float>();
    __hip_bfloat16* hbp =
        reinterpret_cast<__hip_bfloat16*>(h_bf.data_ptr());
    float* owp = o_ws.data_ptr<float>();
    float* hsp = hstats.data_ptr<float>();
    float* csp = cstats.data_ptr<float>();
    int Ti = (int)T, Bi = (int)B, Hi = (int)H;
    void* args[] = {&igp, &h0p, &c0p, &wp2, &lhw, &lhb, &lcw, &lcb,
                    &hap, &cap, &hgp, &crp, &hbp, &owp, &hsp, &csp,
                    &Ti, &Bi, &Hi};
    hipError_t err = hipLaunchCooperativeKernel(
        (const void*)lnlstm_forward_coop_kernel, dim3(blocks), dim3(256),
        args, 0, stream.stream());
    TORCH_CHECK(err == hipSuccess, "coop lnlstm launch: ",
                hipGetErrorString(err));
    return {h_all, c_all, hgates_raw, cellraw};
  }
  hipLaunchKernelGGL(lnlstm_forward_kernel, dim3(B), dim3(kNT), lds,
                     stream.stream(),
                     igates.data_ptr<float>(), h0.data_ptr<float>(),
                     c0.data_ptr<float>(),
                     reinterpret_cast<const __hip_bfloat16*>(w_hh_bf16.data_ptr()),
                     lnh_w.data_ptr<float>(), lnh_b.data_ptr<float>(),
                     lnc_w.data_ptr<float>(), lnc_b.data_ptr<float>(),
                     h_all.data_ptr<float>(), c_all.data_ptr<float>(),
                     hgates_raw.data_ptr<float>(), cellraw.data_ptr<float>(),
                     (int)T, (int)B, (int)H);
  return {h_all, c_all, hgates_raw, cellraw};
}

std::vector<torch::Tensor> lnlstm_backward(
    torch::Tensor dout, torch::Tensor dhT, torch::Tensor dcT,
    torch::Tensor igates, torch::Tensor h_all, torch::Tensor c_all,
    torch::Tensor hgates_raw, torch::Tensor cellraw, torch::Tensor w_hh_t_bf16,
    torch::Tensor lnh_w, torch::Tensor lnh_b, torch::Tensor lnc_w,
    torch::Tensor lnc_b) {
  check_2d(dout, "dout");
  TORCH_CHECK(w_hh_t_bf16.scalar_type() == torch::kBFloat16);
  int64_t T = igates.size(0), B = igates.size(1), G = igates.size(2);
  int64_t H = G / 4;
  auto opt = igates.options();
  auto digates = torch::empty({T, B, G}, opt);
  auto dhgates_raw = torch::empty({T, B, G}, opt);
  auto dh0 = torch::empty({B, H}, opt);
  auto dc0 = torch::empty({B, H}, opt);
  auto dlnh_w = torch::zeros({G}, opt);
  auto dlnh_b = torch::zeros({G}, opt);
  auto dlnc_w = torch::zeros({H}, opt);
  auto dlnc_b = torch::zeros({H}, opt);
  size_t lds = sizeof(float) * (3 * G + 2 * H + kNT);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lnlstm_backward_kernel, dim3(B), dim3(kNT), lds,
                     stream.stream(),
                     dout.data_ptr<float>(),
                     (dhT.defined() && dhT.numel()) ? dhT.data_ptr<float>() : nullptr,
                     (dcT.defined() && dcT.numel()) ? dcT.data_ptr<float>() : nullptr,
                     igates.data_ptr<float>(), h_all.data_ptr<float>(),
                     c_all.data_ptr<float>(), hgates_raw.data_ptr<float>(),
                     cellraw.data_ptr<float>(),
                     reinterpret_cast<const __hip_bfloat16*>(w_hh_t_bf16.data_ptr()),
                     lnh_w.data_ptr<float>(), lnh_b.data_ptr<float>(),
                     lnc_w.data_ptr<float>(), lnc_b.data_ptr<float>(),
                     digates.data_ptr<float>(), dhgates_raw.data_ptr<float>(),
                     dh0.data_ptr<float>(), dc0.data_ptr<float>(),
                     dlnh_w.data_ptr<float>(), dlnh_b.data_ptr<float>(),
                     dlnc_w.data_ptr<float>(), dlnc_b.data_ptr<float>(),
                     (int)T, (int)B, (int)H);
  return {digates, dhgates_raw, dh0, dc0, dlnh_w, dlnh_b, dlnc_w, dlnc_b};
}

extern "C" __global__ void masked_ce_fwd_kernel(
    const float*, const long*, const float*, float*, float*, int, int);
extern "C" __global__ void masked_ce_bwd_kernel(
    const float*, const long*, const float*, const float*, const float*,
    float*, int, int);

extern "C" __global__ void entropy_fwd_kernel(
    const float*, float*, float*, int, int);
extern "C" __global__ void entropy_bwd_kernel(
    const float*, const float*, const float*, const float*, float*, int, int);
extern "C" __global__ void kl_fwd_kernel(
    const float*, const float*, float*, float*, float*, int, int);
extern "C" __global__ void kl_bwd_kernel(
    const float*, const float*, const float*, const float*, const float*,
    float*, int, int);

extern "C" __global__ void su_sample_kernel(
    const float*, const __hip_bfloat16*, const unsigned char*,
    const unsigned char*, const int*, const float*,
    const __hip_bfloat16*, const float*, const __hip_bfloat16*, const float*,
    const __hip_bfloat16*, const __hip_bfloat16*,
    const float*, const float*, const float*, const float*, const float*,
    const float*, const __hip_bfloat16*, const float*, const __hip_bfloat16*,
    const float*, float, float*, int*, int*, int, int, int);

std::vector<torch::Tensor> su_sample(
    torch::Tensor ae_base, torch::Tensor keys, torch::Tensor avail,
    torch::Tensor su_mask, torch::Tensor entity_num, torch::Tensor uniforms,
    torch::Tensor Wq1, torch::Tensor bq1, torch::Tensor Wq2, torch::Tensor bq2,
    torch::Tensor Wih, torch::Tensor Whh,
    torch::Tensor lni_w, torch::Tensor lni_b,
    torch::Tensor lnh_w, torch::Tensor lnh_b,
    torch::Tensor lnc_w, torch::Tensor lnc_b,
    torch::Tensor We1, torch::Tensor be1, torch::Tensor We2, torch::Tensor be2,
    double temperature) {
  int64_t B = ae_base.size(0), AE = ae_base.size(1), N1 = keys.size(1);
  TORCH_CHECK(ae_base.is_cuda() && ae_base.is_contiguous());
  TORCH_CHECK(keys.scalar_type() == torch::kBFloat16 && keys.is_contiguous());
  auto opt = ae_base.options();
  auto logits_out = torch::full({B, 64, N1}, -1e9, opt);
  auto results = torch::zeros({B, 64}, opt.dtype(torch::kInt32));
  auto num_out = torch::zeros({B}, opt.dtype(torch::kInt32));
  size_t lds = sizeof(float) * (AE + 256 + 32 + 128 + 32 + 32 + 32 + 256 + N1 + 256)
      + ((N1 + 63) / 64) * 64 + N1 * 32 * sizeof(__hip_bfloat16);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(su_sample_kernel, dim3(B), dim3(256), lds, stream.stream(),
      ae_base.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(keys.data_ptr()),
      avail.data_ptr<unsigned char>(), su_mask.data_ptr<unsigned char>(),
      entity_num.data_ptr<int>(), uniforms.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(Wq1.data_ptr()), bq1.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(Wq2.data_ptr()), bq2.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(Wih.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(Whh.data_ptr()),
      lni_w.data_ptr<float>(), lni_b.data_ptr<float>(),
      lnh_w.data_ptr<float>(), lnh_b.data_ptr<float>(),
      lnc_w.data_ptr<float>(), lnc_b.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(We1.data_ptr()), be1.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(We2.data_ptr()), be2.data_ptr<float>(),
      (float)temperature,
      logits_out.data_ptr<float>(), results.data_ptr<int>(),
      num_out.data_ptr<int>(), (int)B, (int)N1, (int)AE);
  return {logits_out, results, num_out};
}

extern "C" __global__ void entity_embed_kernel(
    const int*, const float*, const int*, const int*, const int*, const int*,
    __hip_bfloat16*, int, int, int, int, int);

torch::Tensor entity_embed(torch::Tensor int_fields, torch::Tensor float_fields,
                           torch::Tensor kinds, torch::Tensor offsets,
                           torch::Tensor sizes, torch::Tensor src_idx,
                           int64_t out_dim) {
  TORCH_CHECK(int_fields.is_cuda() && int_fields.is_contiguous());
  TORCH_CHECK(int_fields.scalar_type() == torch::kInt32);
  TORCH_CHECK(float_fields.scalar_type() == torch::kFloat32);
  int64_t R = int_fields.size(0);
  int64_t n_int = int_fields.size(1);
  int64_t n_float = float_fields.numel() ? float_fields.size(1) : 0;
  int64_t n_fields = kinds.size(0);
  auto out = torch::empty({R, out_dim},
                          int_fields.options().dtype(torch::kBFloat16));
  int threads = 256;                      // 4 waves -> 4 rows per block
  int blocks = std::min<int64_t>((R + 3) / 4, 8192);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(entity_embed_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(),
                     int_fields.data_ptr<int>(),
                     n_float ? float_fields.data_ptr<float>() : nullptr,
                     kinds.data_ptr<int>(), offsets.data_ptr<int>(),
                     sizes.data_ptr<int>(), src_idx.data_ptr<int>(),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                     (int)R, (int)out_dim, (int)n_fields, (int)n_int,
                     (int)n_float);
  return out;
}

}  // namespace

std::vector<torch::Tensor> masked_ce_fwd(torch::Tensor logits,
                                         torch::Tensor labels,
                                         c10::optional<torch::Tensor> mask) {
  check_2d(logits, "logits");
  TORCH_CHECK(labels.scalar_type() == torch::kLong, "labels must be int64");
  int64_t N = logits.size(0), C = logits.size(1);
  auto loss = torch::empty({N}, logits.options());
  auto lse = torch::empty({N}, logits.options());
  const float* mptr = nullptr;
  if (mask.has_value()) {
    TORCH_CHECK(mask->scalar_type() == torch::kFloat32, "mask must be fp32");
    mptr = mask->data_ptr<float>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(masked_ce_fwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(),
                     logits.data_ptr<float>(), labels.data_ptr<long>(), mptr,
                     loss.data_ptr<float>(), lse.data_ptr<float>(),
                     (int)N, (int)C);
  return {loss, lse};
}

torch::Tensor masked_ce_bwd(torch::Tensor logits, torch::Tensor labels,
                            c10::optional<torch::Tensor> mask,
                            torch::Tensor lse, torch::Tensor gout) {
  check_2d(logits, "logits");
  int64_t N = logits.size(0), C = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const float* mptr = nullptr;
  if (mask.has_value()) mptr = mask->data_ptr<float>();
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(masked_ce_bwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(),
                     logits.data_ptr<float>(), labels.data_ptr<long>(), mptr,
                     lse.data_ptr<float>(), gout.data_ptr<float>(),
                     dlogits.data_ptr<float>(), (int)N, (int)C);
  return dlogits;
}

std::vector<torch::Tensor> entropy_fwd(torch::Tensor logits) {
  check_2d(logits, "logits");
  int64_t N = logits.size(0), C = logits.size(1);
  auto ent = torch::empty({N}, logits.options());
  auto lse = torch::empty({N}, logits.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(entropy_fwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(), logits.data_ptr<float>(),
                     ent.data_ptr<float>(), lse.data_ptr<float>(),
                     (int)N, (int)C);
  return {ent, lse};
}

torch::Tensor entropy_bwd(torch::Tensor logits, torch::Tensor lse,
                          torch::Tensor ent, torch::Tensor gout) {
  check_2d(logits, "logits");
  int64_t N = logits.size(0), C = logits.size(1);
  auto d = torch::empty_like(logits);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(entropy_bwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(), logits.data_ptr<float>(),
                     lse.data_ptr<float>(), ent.data_ptr<float>(),
                     gout.data_ptr<float>(), d.data_ptr<float>(),
                     (int)N, (int)C);
  return d;
}

std::vector<torch::Tensor> kl_fwd(torch::Tensor t_logits, torch::Tensor s_logits) {
  check_2d(t_logits, "t_logits");
  check_2d(s_logits, "s_logits");
  int64_t N = t_logits.size(0), C = t_logits.size(1);
  auto kl = torch::empty({N}, t_logits.options());
  auto t_lse = torch::empty({N}, t_logits.options());
  auto s_lse = torch::empty({N}, t_logits.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(kl_fwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(), t_logits.data_ptr<float>(),
                     s_logits.data_ptr<float>(), kl.data_ptr<float>(),
                     t_lse.data_ptr<float>(), s_lse.data_ptr<float>(),
                     (int)N, (int)C);
  return {kl, t_lse, s_lse};
}

torch::Tensor kl_bwd(torch::Tensor t_logits, torch::Tensor s_logits,
                     torch::Tensor t_lse, torch::Tensor s_lse,
                     torch::Tensor gout) {
  check_2d(s_logits, "s_logits");
  int64_t N = s_logits.size(0), C = s_logits.size(1);
  auto d = torch::empty_like(s_logits);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(kl_bwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     stream.stream(), t_logits.data_ptr<float>(),
                     s_logits.data_ptr<float>(), t_lse.data_ptr<float>(),
                     s_lse.data_ptr<float>(), gout.data_ptr<float>(),
                     d.data_ptr<float>(), (int)N, (int)C);
  return d;
}

// ------------------------------------------------------------- entity attn
extern "C" __global__ void entity_attn_fwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const int*, __hip_bfloat16*, float*, float, int, int, int,
    long, long, long, long, long);
extern "C" __global__ void attn_drow_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*, int, int, int,
    long, long);
extern "C" __global__ void entity_attn_bwd_kv_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const float*, const float*, const int*,
    __hip_bfloat16*, __hip_bfloat16*, float, int, int, int,
    long, long, long, long, long, long, long);
extern "C" __global__ void entity_attn_bwd_q_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const float*, const float*, const int*,
    __hip_bfloat16*, float, int, int, int,
    long, long, long, long, long, long, long);
extern "C" __global__ void mfma_selftest_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*);

static const __hip_bfloat16* bfp(const torch::Tensor& t) {
  return reinterpret_cast<const __hip_bfloat16*>(t.data_ptr());
}
static __hip_bfloat16* bfp_mut(torch::Tensor& t) {
  return reinterpret_cast<__hip_bfloat16*>(t.data_ptr());
}

// qkv: (B, N, 3*H*128) bf16 contiguous (packed [q|k|v][head][dim] order, the
// attention_pre fc output); entity_num: (B,) int32 or None.
std::vector<torch::Tensor> entity_attn_fwd(
    torch::Tensor qkv, c10::optional<torch::Tensor> entity_num,
    int64_t H, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous() && qkv.dim() == 3);
  TORCH_CHECK(qkv.scalar_type() == torch::kBFloat16, "qkv must be bf16");
  int64_t B = qkv.size(0), N = qkv.size(1), F = qkv.size(2);
  int64_t HD = F / 3;
  TORCH_CHECK(HD == H * 128, "entity_attn: head_dim must be 128");
  auto out = torch::empty({B, N, HD}, qkv.options());
  auto lse = torch::empty({B, H, N}, qkv.options().dtype(torch::kFloat32));
  const int* en = nullptr;
  if (entity_num.has_value()) {
    TORCH_CHECK(entity_num->scalar_type() == torch::kInt32);
    en = entity_num->data_ptr<int>();
  }
  dim3 grid((N + 63) / 64, H, B);
  size_t lds = 64 * 256 * 2 + 128 * 128 + 64 * 128;    // Q,K,Vt,Ps
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(entity_attn_fwd_kernel, grid, dim3(256), lds,
                     stream.stream(),
                     bfp(qkv), bfp(qkv) + HD, bfp(qkv) + 2 * HD, en,
                     bfp_mut(out), lse.data_ptr<float>(), (float)scale,
                     (int)B, (int)H, (int)N,
                     (long)(N * F), (long)F, 128L,
                     (long)(N * HD), (long)HD);
  return {out, lse};
}

torch::Tensor entity_attn_bwd(
    torch::Tensor qkv, c10::optional<torch::Tensor> entity_num,
    torch::Tensor out, torch::Tensor dout, torch::Tensor lse,
    int64_t H, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous());
  TORCH_CHECK(dout.is_contiguous() && dout.scalar_type() == torch::kBFloat16);
  int64_t B = qkv.size(0), N = qkv.size(1), F = qkv.size(2);
  int64_t HD = F / 3;
  auto drow = torch::empty({B, H, N}, lse.options());
  auto dqkv = torch::empty_like(qkv);
  const int* en = nullptr;
  if (entity_num.has_value()) en = entity_num->data_ptr<int>();
  auto stream = c10::hip::getCurrentHIPStream();
  long total_rows = B * H * N;
  int drow_blocks = (int)std::min<long>((total_rows + 3) / 4, 8192);
  hipLaunchKernelGGL(attn_drow_kernel, dim3(drow_blocks), dim3(256), 0,
                     stream.stream(), bfp(dout), bfp(out),
                     drow.data_ptr<float>(), (int)B, (int)H, (int)N,
                     (long)(N * HD), (long)HD);
  dim3 grid((N + 63) / 64, H, B);
  size_t lds_kv = 4 * 64 * 256 + 64 * 128 + 2 * 64 * sizeof(float);
  hipLaunchKernelGGL(entity_attn_bwd_kv_kernel, grid, dim3(256), lds_kv,
                     stream.stream(),
                     bfp(qkv), bfp(qkv) + HD, bfp(qkv) + 2 * HD,
                     bfp(dout), lse.data_ptr<float>(), drow.data_ptr<float>(),
                     en, bfp_mut(dqkv) + HD, bfp_mut(dqkv) + 2 * HD,
                     (float)scale, (int)B, (int)H, (int)N,
                     (long)(N * F), (long)F, 128L,
                     (long)(N * HD), (long)HD,
                     (long)(N * F), (long)F);
  size_t lds_q = 4 * 64 * 256 + 64 * 128 + 2 * 64 * sizeof(float);
  hipLaunchKernelGGL(entity_attn_bwd_q_kernel, grid, dim3(256), lds_q,
                     stream.stream(),
                     bfp(qkv), bfp(qkv) + HD, bfp(qkv) + 2 * HD,
                     bfp(dout), lse.data_ptr<float>(), drow.data_ptr<float>(),
                     en, bfp_mut(dqkv),
                     (float)scale, (int)B, (int)H, (int)N,
                     (long)(N * F), (long)F, 128L,
                     (long)(N * HD), (long)HD,
                     (long)(N * F), (long)F);
  return dqkv;
}

extern "C" __global__ void scatter_add_bf16_kernel(
    const __hip_bfloat16*, const int*, const int*, __hip_bfloat16*,
    int, int, int, int, int);
extern "C" __global__ void scatter_add_bf16_bwd_kernel(
    const __hip_bfloat16*, const int*, __hip_bfloat16*,
    int, int, int, int, int);

torch::Tensor scatter_add_map(torch::Tensor src, torch::Tensor xy,
                              c10::optional<torch::Tensor> entity_num,
                              int64_t H, int64_t W) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous() && src.dim() == 3);
  TORCH_CHECK(src.scalar_type() == torch::kBFloat16, "src must be bf16");
  TORCH_CHECK(xy.scalar_type() == torch::kInt32 && xy.is_contiguous());
  int64_t B = src.size(0), N = src.size(1), C = src.size(2);
  auto out = torch::zeros({B, C, H, W}, src.options());
  const int* en = nullptr;
  if (entity_num.has_value()) en = entity_num->data_ptr<int>();
  long total = B * N * C;
  int blocks = (int)std::min<long>((total + 255) / 256, 4096);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(scatter_add_bf16_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), bfp(src), xy.data_ptr<int>(), en,
                     bfp_mut(out), (int)B, (int)N, (int)C, (int)H, (int)W);
  return out;
}

torch::Tensor scatter_add_map_bwd(torch::Tensor dout, torch::Tensor xy,
                                  int64_t N) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && dout.dim() == 4);
  TORCH_CHECK(dout.scalar_type() == torch::kBFloat16);
  int64_t B = dout.size(0), C = dout.size(1), H = dout.size(2), W = dout.size(3);
  auto dsrc = torch::empty({B, N, C}, dout.options());
  long total = B * N * C;
  int blocks = (int)std::min<long>((total + 255) / 256, 4096);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(scatter_add_bf16_bwd_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), bfp(dout), xy.data_ptr<int>(),
                     bfp_mut(dsrc), (int)B, (int)N, (int)C, (int)H, (int)W);
  return dsrc;
}

extern "C" __global__ void residual_ln_fwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*, const float*,
    __hip_bfloat16*, __hip_bfloat16*, float*, float*, long, int, float);
extern "C" __global__ void residual_ln_bwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*, const float*,
    const float*, __hip_bfloat16*, float*, float*, long, int);

std::vector<torch::Tensor> residual_ln_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> a, torch::Tensor w,
    torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  int64_t C = x.size(-1);
  TORCH_CHECK(C == 256, "residual_ln kernel is specialized to C=256");
  long R = x.numel() / C;
  auto y = torch::empty_like(x);
  auto s = torch::empty_like(x);
  auto fopt = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({R}, fopt);
  auto rstd = torch::empty({R}, fopt);
  const __hip_bfloat16* ap = nullptr;
  if (a.has_value()) {
    TORCH_CHECK(a->is_contiguous() && a->scalar_type() == torch::kBFloat16);
    ap = bfp(*a);
  }
  int blocks = (int)std::min<long>((R + 3) / 4, 1 << 20);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(residual_ln_fwd_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), bfp(x), ap,
                     w.data_ptr<float>(), b.data_ptr<float>(),
                     bfp_mut(y), bfp_mut(s), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), R, (int)C, (float)eps);
  return {y, s, mean, rstd};
}

std::vector<torch::Tensor> residual_ln_bwd(
    torch::Tensor dy, torch::Tensor s, torch::Tensor mean,
    torch::Tensor rstd, torch::Tensor w) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16);
  int64_t C = dy.size(-1);
  long R = dy.numel() / C;
  auto dsum = torch::empty_like(dy);
  auto fopt = dy.options().dtype(torch::kFloat32);
  auto dw = torch::zeros({C}, fopt);
  auto db = torch::zeros({C}, fopt);
  int blocks = (int)std::min<long>((R + 3) / 4, 8192);
  size_t lds = 8 * C * sizeof(float);    // [4 waves][2*C] combine buffer
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(residual_ln_bwd_kernel, dim3(blocks), dim3(256), lds,
                     stream.stream(), bfp(dy), bfp(s),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     w.data_ptr<float>(), bfp_mut(dsum),
                     dw.data_ptr<float>(), db.data_ptr<float>(), R, (int)C);
  return {dsum, dw, db};
}

extern "C" __global__ void conv2d_fwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, int, int, int, int, int, int, int, int, int, int, int);
extern "C" __global__ void conv2d_small_fwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, int, int, int, int, int, int, int, int, int, int, int);
extern "C" __global__ void conv2d_wgrad_small_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*,
    int, int, int, int, int, int, int, int, int, int);
extern "C" __global__ void conv2d_fwd_smallhw_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, int, int, int, int, int, int, int);
extern "C" __global__ void conv2d_wgrad_smallhw_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*, float*,
    int, int, int, int, int, int, int);
extern "C" __global__ void im2col_3x3_kernel(
    const __hip_bfloat16*, __hip_bfloat16*, long, int, int);
extern "C" __global__ void col2im_3x3_kernel(
    const __hip_bfloat16*, __hip_bfloat16*, long, int, int);
extern "C" __global__ void conv2d_stencil_c1_fwd_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, int, int, int, int, int, int);
extern "C" __global__ void conv2d_stencil_c1_wgrad_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*, float*,
    int, int, int, int);
extern "C" __global__ void conv2d_wgrad_kernel(
    const __hip_bfloat16*, const __hip_bfloat16*, float*, float*,
    int, int, int, int, int, int, int, int, int, int, int);
extern "C" __global__ void maxpool2x2_fwd_kernel(
    const __hip_bfloat16*, __hip_bfloat16*, unsigned char*, long, int, int);
extern "C" __global__ void maxpool2x2_bwd_kernel(
    const __hip_bfloat16*, const unsigned char*, __hip_bfloat16*,
    long, int, int);

torch::Tensor conv2d_fwd(torch::Tensor input, torch::Tensor wp,
                         c10::optional<torch::Tensor> bias,
                         int64_t Cout, int64_t KH, int64_t KW,
                         int64_t padH, int64_t padW, bool relu) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous() && input.dim() == 4);
  TORCH_CHECK(input.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(wp.is_contiguous() && wp.scalar_type() == torch::kBFloat16);
  int64_t B = input.size(0), Cin = input.size(1);
  int64_t H = input.size(2), W = input.size(3);
  int64_t Kpad = wp.size(1);
  auto out = torch::empty({B, Cout, H, W}, input.options());
  const float* bp = nullptr;
  if (bias.has_value()) bp = bias->data_ptr<float>();
  int HW = (int)(H * W);
  auto stream = c10::hip::getCurrentHIPStream();
  if (Cout == 1 && KH == 3 && KW == 3 && padH == 1 && padW == 1 &&
      Cin % 16 == 0 && Cin <= 256 && B < 65536) {
    // direct-load stencil: row items, vector loads, register partials
    long items = B * H * ((W + 15) / 16);
    int sgb = (int)std::min<long>((items + 255) / 256, 8192);
    hipLaunchKernelGGL(conv2d_stencil_c1_fwd_kernel, dim3(sgb), dim3(256), 0,
                       stream.stream(), bfp(input), bfp(wp), bp,
                       bfp_mut(out), (int)B, (int)Cin, (int)H, (int)W,
                       (int)Kpad, relu ? 1 : 0);
    return out;
  }
  if (Cout < 16) {
    // memory-bound tiny-channel conv (e.g. the 32->1 location head):
    // direct VALU kernel, one thread per output pixel
    long total = B * Cout * (long)HW;
    int blocks = (int)std::min<long>((total + 255) / 256, 8192);
    hipLaunchKernelGGL(conv2d_small_fwd_kernel, dim3(blocks), dim3(256), 0,
                       stream.stream(), bfp(input), bfp(wp), bp,
                       bfp_mut(out), (int)B, (int)Cin, (int)Cout,
                       (int)H, (int)W, (int)KH, (int)KW, (int)padH,
                       (int)padW, (int)Kpad, relu ? 1 : 0);
    return out;
  }
  dim3 grid((HW + 63) / 64, (unsigned)((Cout + 127) / 128), (unsigned)B);
  // (a windowed fwd variant was measured SLOWER: each block owns one
  // 64-px tile, so the per-chunk window restage gets no cross-tile reuse
  // — see profiles/r02_notes.md; the wgrad variant below keeps it because
  // its px loop is INSIDE the block)
  hipLaunchKernelGGL(conv2d_fwd_kernel, grid, dim3(256), 64 * 256,
                     stream.stream(), bfp(input), bfp(wp), bp, bfp_mut(out),
                     (int)B, (int)Cin, (int)Cout, (int)H, (int)W,
                     (int)KH, (int)KW, (int)padH, (int)padW, (int)Kpad,
                     relu ? 1 : 0);
  return out;
}

std::vector<torch::Tensor> conv2d_wgrad(
    torch::Tensor input, torch::Tensor dout,
    int64_t KH, int64_t KW, int64_t padH, int64_t padW,
    int64_t Kpad, bool want_bias) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous());
  TORCH_CHECK(dout.is_contiguous() && dout.scalar_type() == torch::kBFloat16);
  int64_t B = input.size(0), Cin = input.size(1);
  int64_t H = input.size(2), W = input.size(3);
  int64_t Cout = dout.size(1);
  auto dwp = torch::zeros({Kpad, Cout},
                          input.options().dtype(torch::kFloat32));
  // pick images-per-block so total blocks lands in a healthy range
  int64_t K_real = Cin * KH * KW;
  if (Cout <= 4) {
    auto dwd = torch::zeros({Cout, Cin, KH, KW},
                            input.options().dtype(torch::kFloat32));
    auto stream2 = c10::hip::getCurrentHIPStream();
    torch::Tensor dbs;
    if (Cout == 1 && KH == 3 && KW == 3 && padH == 1 && padW == 1 &&
        Cin <= 32 && (Cin & (Cin - 1)) == 0) {
      // direct-load stencil wgrad with fused dbias (LocationHead 32->1)
      float* dbp2 = nullptr;
      if (want_bias) {
        dbs = torch::zeros({1}, input.options().dtype(torch::kFloat32));
        dbp2 = dbs.data_ptr<float>();
      }
      long items = B * H * Cin;
      int g = (int)std::min<long>((items + 255) / 256, 8192);
      hipLaunchKernelGGL(conv2d_stencil_c1_wgrad_kernel, dim3(g), dim3(256),
                         0, stream2.stream(), bfp(input), bfp(dout),
                         dwd.data_ptr<float>(), dbp2,
                         (int)B, (int)Cin, (int)H, (int)W);
    } else {
      // tiny-Cout direct wgrad: dW (Cout, Cin, KH, KW) fp32, one input
      // read per block, per-thread register partials (see conv2d.hip)
      int ipb2 = (int)std::max<int64_t>(
          1, B / std::max<int64_t>(1, 4096 / Cin));
      dim3 g2((unsigned)Cin, (unsigned)((B + ipb2 - 1) / ipb2));
      hipLaunchKernelGGL(conv2d_wgrad_small_kernel, g2, dim3(256), 0,
                         stream2.stream(), bfp(input), bfp(dout),
                         dwd.data_ptr<float>(), (int)B, (int)Cin, (int)Cout,
                         (int)H, (int)W, (int)KH, (int)KW, (int)padH,
                         (int)padW, ipb2);
      if (want_bias)
        dbs = dout.sum(torch::IntArrayRef{0, 2, 3}, false, torch::kFloat32);
    }
    // pack into the (Kpad, Cout) layout the python wrapper slices
    auto dwp = torch::zeros({Kpad, Cout},
                            input.options().dtype(torch::kFloat32));
    dwp.narrow(0, 0, K_real).copy_(
        dwd.permute({1, 2, 3, 0}).reshape({K_real, Cout}));
    return {dwp, dbs};
  }
  int kt = (int)((std::min<int64_t>(Kpad, (K_real + 31) / 32 * 32) + 63) / 64);
  long want_z = 32768 / std::max(1, kt);
  int ipb = (int)std::max<long>(1, (B + want_z - 1) / std::max<long>(1, want_z));
  // small images: batch enough px-chunks per block that the per-block
  // dW atomic flush amortizes (at 19x20 the old heuristic gave 1024
  // image-blocks per k-tile = 1024 serialized atomic adds per dW cell)
  long px_chunks = (H * W + 127) / 128;
  ipb = std::max<long>(ipb, std::min<int64_t>(B, 24 / std::max<long>(1, px_chunks)));
  dim3 grid(kt, 1, (unsigned)((B + ipb - 1) / ipb));
  torch::Tensor dbias;
  float* dbp = nullptr;
  if (want_bias) {
    dbias = torch::zeros({Cout}, input.options().dtype(torch::kFloat32));
    dbp = dbias.data_ptr<float>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  if (KH == 3 && KW == 3 && padH == 1 && padW == 1 &&
      (H + 2) * (W + 2) <= 484 && W >= 8) {
    hipLaunchKernelGGL(conv2d_wgrad_smallhw_kernel, grid, dim3(256),
                       64 * 256, stream.stream(), bfp(input), bfp(dout),
                       dwp.data_ptr<float>(), dbp,
                       (int)B, (int)Cin, (int)Cout, (int)H, (int)W,
                       (int)Kpad, ipb);
    return {dwp, dbias};
  }
  hipLaunchKernelGGL(conv2d_wgrad_kernel, grid, dim3(256), 64 * 256,
                     stream.stream(), bfp(input), bfp(dout),
                     dwp.data_ptr<float>(), dbp,
                     (int)B, (int)Cin, (int)Cout, (int)H, (int)W,
                     (int)KH, (int)KW, (int)padH, (int)padW, (int)Kpad, ipb);
  return {dwp, dbias};
}

torch::Tensor im2col3x3(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  int64_t B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto col = torch::empty({B, C * 9, H * W}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(im2col_3x3_kernel, dim3((unsigned)(B * C), 9),
                     dim3(256), 0, stream.stream(), bfp(x), bfp_mut(col),
                     (long)(B * C), (int)H, (int)W);
  return col;
}

torch::Tensor col2im3x3(torch::Tensor dcol, int64_t H, int64_t W) {
  TORCH_CHECK(dcol.is_cuda() && dcol.is_contiguous() && dcol.dim() == 3);
  TORCH_CHECK(dcol.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dcol.size(1) % 9 == 0 && dcol.size(2) == H * W);
  int64_t B = dcol.size(0), C = dcol.size(1) / 9;
  auto dx = torch::empty({B, C, H, W}, dcol.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(col2im_3x3_kernel, dim3((unsigned)(B * C)),
                     dim3(256), 0, stream.stream(), bfp(dcol), bfp_mut(dx),
                     (long)(B * C), (int)H, (int)W);
  return dx;
}

std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor input) {
  TORCH_CHECK(input.is_cuda() && input.is_contiguous() && input.dim() == 4);
  TORCH_CHECK(input.scalar_type() == torch::kBFloat16);
  int64_t B = input.size(0), C = input.size(1);
  int64_t H = input.size(2), W = input.size(3);
  auto out = torch::empty({B, C, H / 2, W / 2}, input.options());
  auto idx = torch::empty({B, C, H / 2, W / 2},
                          input.options().dtype(torch::kUInt8));
  long total = out.numel();
  int blocks = (int)std::min<long>((total + 255) / 256, 4096);
  auto stream = c10::hip::getCurrentHIPStream();
  if (W % 16 == 0) {
    long groups = B * C * (H / 2) * (W / 16);
    int gb = (int)std::min<long>((groups + 255) / 256, 4096);
    hipLaunchKernelGGL(maxpool2x2_fwd_vec_kernel, dim3(gb), dim3(256), 0,
                       stream.stream(), bfp(input), bfp_mut(out),
                       idx.data_ptr<unsigned char>(), (long)(B * C),
                       (int)H, (int)W);
  } else {
    hipLaunchKernelGGL(maxpool2x2_fwd_kernel, dim3(blocks), dim3(256), 0,
                       stream.stream(), bfp(input), bfp_mut(out),
                       idx.data_ptr<unsigned char>(), (long)(B * C),
                       (int)H, (int)W);
  }
  return {out, idx};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor dout, torch::Tensor idx,
                             int64_t H, int64_t W) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous());
  int64_t B = dout.size(0), C = dout.size(1);
  auto din = torch::empty({B, C, H, W}, dout.options());
  long total = din.numel();
  int blocks = (int)std::min<long>((total + 255) / 256, 4096);
  auto stream = c10::hip::getCurrentHIPStream();
  if (W % 8 == 0) {
    long groups = B * C * H * (W / 8);
    int gb = (int)std::min<long>((groups + 255) / 256, 4096);
    hipLaunchKernelGGL(maxpool2x2_bwd_vec_kernel, dim3(gb), dim3(256), 0,
                       stream.stream(), bfp(dout),
                       idx.data_ptr<unsigned char>(), bfp_mut(din),
                       (long)(B * C), (int)H, (int)W);
  } else {
    hipLaunchKernelGGL(maxpool2x2_bwd_kernel, dim3(blocks), dim3(256), 0,
                       stream.stream(), bfp(dout),
                       idx.data_ptr<unsigned char>(), bfp_mut(din),
                       (long)(B * C), (int)H, (int)W);
  }
  return din;
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor Bm) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous() && Bm.is_contiguous());
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32);
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0,
                     stream.stream(), bfp(A), bfp(Bm), D.data_ptr<float>());
  return D;
}

extern "C" __global__ void multi_norm_sq_kernel(
    const unsigned long long*, const long*, int, float*);
extern "C" __global__ void multi_scale_kernel(
    const unsigned long long*, const long*, int, const float*,
    float, float, int);

static std::pair<torch::Tensor, int> build_chunk_table(
    const std::vector<torch::Tensor>& grads) {
  int n = (int)grads.size();
  auto meta = torch::empty({2, n}, torch::dtype(torch::kInt64));
  auto acc = meta.accessor<int64_t, 2>();
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(grads[i].is_cuda() && grads[i].is_contiguous());
    TORCH_CHECK(grads[i].scalar_type() == torch::kFloat32);
    acc[0][i] = (int64_t)grads[i].data_ptr();
    acc[1][i] = grads[i].numel();
  }
  return {meta.to(grads[0].device(), /*non_blocking=*/true), n};
}

torch::Tensor multi_norm_sq(std::vector<torch::Tensor> grads) {
  TORCH_CHECK(!grads.empty());
  auto [meta, n] = build_chunk_table(grads);
  auto out = torch::zeros({}, grads[0].options());
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = std::min(n, 2048);
  hipLaunchKernelGGL(multi_norm_sq_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const unsigned long long*>(
                         meta[0].data_ptr<int64_t>()),
                     meta[1].data_ptr<int64_t>(), n, out.data_ptr<float>());
  return out;
}

void multi_clip(std::vector<torch::Tensor> grads, torch::Tensor norm_sq,
                double thresh, double eps) {
  auto [meta, n] = build_chunk_table(grads);
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(64, std::min(n, 1024));
  hipLaunchKernelGGL(multi_scale_kernel, grid, dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const unsigned long long*>(
                         meta[0].data_ptr<int64_t>()),
                     meta[1].data_ptr<int64_t>(), n,
                     norm_sq.data_ptr<float>(), (float)thresh, (float)eps, 1);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("multi_norm_sq", &multi_norm_sq,
        "K14 multi-tensor global sum-of-squares");
  m.def("multi_clip", &multi_clip,
        "K14 clip-by-global-norm with a device-resident norm (no host sync)");
  m.def("entity_attn_fwd", &entity_attn_fwd,
        "K1 entity-transformer flash attention forward (bf16 MFMA)");
  m.def("entity_attn_bwd", &entity_attn_bwd,
        "K1 entity-transformer flash attention backward");
  m.def("mfma_selftest", &mfma_selftest,
        "one 16x16x32 bf16 MFMA: A(16,32) . B(16,32)^T -> (16,16) fp32");
  m.def("scatter_add_map", &scatter_add_map,
        "K3 entity->NCHW map scatter-add (packed bf16 atomics)");
  m.def("scatter_add_map_bwd", &scatter_add_map_bwd,
        "K3 scatter-add backward (row gather)");
  m.def("residual_ln_fwd", &residual_ln_fwd,
        "fused residual-add + LayerNorm forward (bf16, fp32 stats)");
  m.def("residual_ln_bwd", &residual_ln_bwd,
        "fused residual-add + LayerNorm backward");
  m.def("conv2d_fwd", &conv2d_fwd,
        "K4 NCHW bf16 MFMA implicit-GEMM conv (stride 1), fused bias/relu");
  m.def("conv2d_wgrad", &conv2d_wgrad, "K4 conv weight gradient");
  m.def("im2col3x3", &im2col3x3, "3x3 pad-1 im2col (B,C,HW)->(B,C*9,HW)");
  m.def("col2im3x3", &col2im3x3, "3x3 pad-1 col2im gather (no atomics)");
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd, "2x2 maxpool fwd + argmax");
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd, "2x2 maxpool gather backward");
  m.def("entropy_fwd", &entropy_fwd, "fused rowwise entropy forward");
  m.def("entropy_bwd", &entropy_bwd, "fused rowwise entropy backward");
  m.def("kl_fwd", &kl_fwd, "fused rowwise KL(teacher||student) forward");
  m.def("kl_bwd", &kl_bwd, "fused rowwise KL backward (student grads)");
  m.def("masked_ce_fwd", &masked_ce_fwd,
        "fused masked CE forward (per-row loss + lse)");
  m.def("masked_ce_bwd", &masked_ce_bwd, "fused masked CE backward");
  m.def("su_sample", &su_sample, "selected-units sampling loop (one kernel)");
  m.def("entity_embed", &entity_embed,
        "fused 36-field entity embedding -> (R, 997) bf16");
  m.def("lambda_return_scan", &lambda_return_scan,
        "generalized lambda-return reverse scan (T,B)");
  m.def("vtrace_scan", &vtrace_scan, "v-trace corrected-value reverse scan");
  m.def("lnlstm_forward", &lnlstm_forward, "fused LN-LSTM layer forward");
  m.def("lnlstm_backward", &lnlstm_backward, "fused LN-LSTM layer backward");
  m.def("upsample2x", &upsample2x, "bilinear 2x upsample (align_corners=false)");
  m.def("upsample2x_backward", &upsample2x_backward, "bilinear 2x upsample backward");
}
