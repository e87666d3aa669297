// PyTorch bindings for the AudioMuse-AMD native kernel library (gfx950).
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

#include <c10/hip/HIPStream.h>

namespace audiomuse {
void launch_mel_fwd(const float* audio, float* out, const float* window,
                    const float2* twiddle, const int* mel_rowptr,
                    const int* mel_bin, const float* mel_w, int B, int T,
                    int n_frames, int hop, int n_mels, int n_fft, int center,
                    int log_mode, int quant16, hipStream_t stream);
void launch_ivf_scan(int dtype_code, int metric, const void* query,
                     const float* qnorm, const void* data,
                     const float* row_norm, const int* probe,
                     const int* cell_off, const long long* cand_off,
                     float* out_dist, int* out_row, int Q, int d, int nprobe,
                     long long cap, hipStream_t stream);
void launch_layernorm_bf16(const void* x, void* y, const void* w,
                           const void* b, long long n_rows, int dim, float eps,
                           hipStream_t stream);
void launch_add_layernorm_bf16(const void* x, const void* in2, void* sum_out,
                               void* y, const void* w, const void* b,
                               long long n_rows, int dim, float eps,
                               hipStream_t stream);
void launch_layernorm_bf16_fp8_impl(const void* x, void* y8, const void* w,
                                    const void* b, const void* in2,
                                    void* sum_out, const float* q_scale,
                                    float* q_amax, long long n_rows, int dim,
                                    float eps, hipStream_t stream);
void launch_mfma_probe(const void* A, const void* B, float* D,
                       hipStream_t stream);
void launch_window_attn4(const void* qkv, void* out, const void* bias,
                         int Bn, int H, int W, int C, int heads, int shift,
                         float scale, hipStream_t stream);
void launch_window_attn(const void* qkv, void* out, const void* bias, int Bn,
                        int H, int W, int C, int heads, int shift, float scale,
                        hipStream_t stream);
void launch_window_attn_fp8(const void* qkv, void* out, const void* bias,
                            const void* qs_ptr, int Bn, int H, int W, int C,
                            int heads, int shift, float sm_scale,
                            hipStream_t stream);
}

#define AM_CHECK(x, msg) TORCH_CHECK(x, msg)
#define AM_CHECK_GPU_F32_CONTIG(t)                                   \
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kFloat &&        \
                  t.is_contiguous(),                                 \
              #t " must be a contiguous float32 GPU tensor")

static torch::Tensor mel_fwd(torch::Tensor audio, torch::Tensor window,
                             torch::Tensor twiddle, torch::Tensor mel_rowptr,
                             torch::Tensor mel_bin, torch::Tensor mel_w,
                             int64_t hop, int64_t n_fft, bool center,
                             int64_t log_mode, bool quant16) {
  AM_CHECK_GPU_F32_CONTIG(audio);
  AM_CHECK_GPU_F32_CONTIG(window);
  AM_CHECK_GPU_F32_CONTIG(mel_w);
  AM_CHECK(audio.dim() == 2, "audio must be (B, T)");
  AM_CHECK(twiddle.is_cuda() && twiddle.is_contiguous() &&
               twiddle.scalar_type() == at::kFloat &&
               twiddle.numel() == n_fft,  // (n_fft/2, 2) floats
           "twiddle must be (n_fft/2, 2) float32 on GPU");
  AM_CHECK(mel_rowptr.is_cuda() && mel_rowptr.scalar_type() == at::kInt &&
               mel_bin.scalar_type() == at::kInt,
           "CSR index tensors must be int32 on GPU");
  AM_CHECK(n_fft == 256 || n_fft == 512 || n_fft == 1024 || n_fft == 2048 ||
               n_fft == 4096,
           "n_fft must be a power of two in [256, 4096]");

  const int64_t B = audio.size(0);
  const int64_t T = audio.size(1);
  const int64_t n_mels = mel_rowptr.numel() - 1;
  const int64_t n_frames =
      center ? (1 + T / hop) : (1 + (T - n_fft) / hop);
  AM_CHECK(n_frames >= 1, "audio too short for one frame");

  auto out = torch::empty({B, n_mels, n_frames}, audio.options());
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_mel_fwd(
      audio.data_ptr<float>(), out.data_ptr<float>(), window.data_ptr<float>(),
      reinterpret_cast<const float2*>(twiddle.data_ptr<float>()),
      mel_rowptr.data_ptr<int>(), mel_bin.data_ptr<int>(),
      mel_w.data_ptr<float>(), (int)B, (int)T, (int)n_frames, (int)hop,
      (int)n_mels, (int)n_fft, center ? 1 : 0, (int)log_mode, quant16 ? 1 : 0,
      stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

static void ivf_scan(int64_t dtype_code, int64_t metric, torch::Tensor query,
                     torch::Tensor qnorm, torch::Tensor data,
                     torch::Tensor row_norm, torch::Tensor probe,
                     torch::Tensor cell_off, torch::Tensor cand_off,
                     torch::Tensor out_dist, torch::Tensor out_row,
                     int64_t dim_pad) {
  AM_CHECK(query.is_cuda() && data.is_cuda(), "ivf_scan needs GPU tensors");
  AM_CHECK(probe.scalar_type() == at::kInt && cell_off.scalar_type() == at::kInt,
           "probe/cell_off must be int32");
  AM_CHECK(cand_off.scalar_type() == at::kLong, "cand_off must be int64");
  AM_CHECK(out_dist.is_contiguous() && out_row.is_contiguous(),
           "outputs must be contiguous");
  const int Q = probe.size(0);
  const int nprobe = probe.size(1);
  const long long cap = out_dist.size(1);
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_ivf_scan(
      (int)dtype_code, (int)metric, query.data_ptr(), qnorm.data_ptr<float>(),
      data.data_ptr(), row_norm.data_ptr<float>(), probe.data_ptr<int>(),
      cell_off.data_ptr<int>(),
      reinterpret_cast<const long long*>(cand_off.data_ptr<int64_t>()),
      out_dist.data_ptr<float>(), out_row.data_ptr<int>(), Q, (int)dim_pad,
      nprobe, cap, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
}

static torch::Tensor layernorm_bf16(torch::Tensor x, torch::Tensor w,
                                    torch::Tensor b, double eps) {
  AM_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
           "x must be contiguous bf16 on GPU");
  const int dim = x.size(-1);
  AM_CHECK(dim % 4 == 0 && dim <= 4096,
           "dim must be a multiple of 4 and <= 4096");
  AM_CHECK(w.is_contiguous() && b.is_contiguous() &&
               w.scalar_type() == at::kBFloat16 &&
               b.scalar_type() == at::kBFloat16 && w.numel() == dim &&
               b.numel() == dim,
           "w/b must be contiguous bf16 of size dim");
  auto y = torch::empty_like(x);
  const long long n_rows = x.numel() / dim;
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_layernorm_bf16(x.data_ptr(), y.data_ptr(), w.data_ptr(),
                                   b.data_ptr(), n_rows, dim, (float)eps,
                                   stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return y;
}

static torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  AM_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
               A.is_contiguous() && A.size(0) == 16 && A.size(1) == 32,
           "A must be (16,32) bf16 GPU");
  AM_CHECK(B.is_cuda() && B.scalar_type() == at::kBFloat16 &&
               B.is_contiguous() && B.size(0) == 32 && B.size(1) == 16,
           "B must be (32,16) bf16 GPU");
  auto D = torch::empty({16, 16}, A.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_mfma_probe(A.data_ptr(), B.data_ptr(),
                               D.data_ptr<float>(), stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return D;
}

static torch::Tensor window_attn_fwd(torch::Tensor qkv, torch::Tensor bias,
                                     int64_t heads, int64_t shift,
                                     double scale) {
  AM_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kBFloat16 &&
               qkv.is_contiguous() && qkv.dim() == 4,
           "qkv must be (B, H, W, 3C) bf16 contiguous GPU");
  const int64_t Bn = qkv.size(0), H = qkv.size(1), W = qkv.size(2);
  const int64_t C = qkv.size(3) / 3;
  AM_CHECK(qkv.size(3) == 3 * C && C == heads * 32,
           "C must be heads*32 and last dim 3C");
  AM_CHECK(H % 8 == 0 && W % 8 == 0, "H, W must be multiples of 8");
  AM_CHECK(heads % 2 == 0, "heads must be a multiple of 2");
  AM_CHECK(bias.is_cuda() && bias.scalar_type() == at::kBFloat16 &&
               bias.is_contiguous() && bias.numel() == heads * 64 * 64,
           "bias must be (heads, 64, 64) bf16 contiguous");
  auto out = torch::empty({Bn, H, W, C}, qkv.options());
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_window_attn(qkv.data_ptr(), out.data_ptr(),
                                bias.data_ptr(), (int)Bn, (int)H,
                                (int)W, (int)C, (int)heads, (int)shift,
                                (float)scale, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

static torch::Tensor window_attn_fp8_fwd(torch::Tensor qkv,
                                         torch::Tensor bias,
                                         torch::Tensor q_scale,
                                         int64_t heads, int64_t shift,
                                         double sm_scale) {
  AM_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kFloat8_e4m3fn &&
               qkv.is_contiguous() && qkv.dim() == 4,
           "qkv must be (B, H, W, 3C) fp8e4m3 contiguous GPU");
  const int64_t Bn = qkv.size(0), H = qkv.size(1), W = qkv.size(2);
  const int64_t C = qkv.size(3) / 3;
  AM_CHECK(qkv.size(3) == 3 * C && C == heads * 32,
           "C must be heads*32 and last dim 3C");
  AM_CHECK(H % 8 == 0 && W % 8 == 0, "H, W must be multiples of 8");
  AM_CHECK(bias.is_cuda() && bias.scalar_type() == at::kBFloat16 &&
               bias.is_contiguous() && bias.numel() == heads * 64 * 64,
           "bias must be (heads, 64, 64) bf16 contiguous");
  AM_CHECK(q_scale.is_cuda() && q_scale.scalar_type() == at::kFloat &&
               q_scale.numel() == 1,
           "q_scale must be a f32 device scalar");
  auto out = torch::empty({Bn, H, W, C},
                          qkv.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_window_attn_fp8(
      qkv.data_ptr(), out.data_ptr(), bias.data_ptr(), q_scale.data_ptr(),
      (int)Bn, (int)H, (int)W, (int)C, (int)heads, (int)shift,
      (float)sm_scale, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

static torch::Tensor window_attn4_fwd(torch::Tensor qkv, torch::Tensor bias,
                                      int64_t heads, int64_t shift,
                                      double scale) {
  AM_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kBFloat16 &&
               qkv.is_contiguous() && qkv.dim() == 4,
           "qkv must be (B, H, W, 3C) bf16 contiguous GPU");
  const int64_t Bn = qkv.size(0), H = qkv.size(1), W = qkv.size(2);
  const int64_t C = qkv.size(3) / 3;
  AM_CHECK(qkv.size(3) == 3 * C && C == heads * 32,
           "C must be heads*32 and last dim 3C");
  AM_CHECK(H % 4 == 0 && W % 4 == 0, "H, W must be multiples of 4");
  AM_CHECK(bias.is_cuda() && bias.scalar_type() == at::kBFloat16 &&
               bias.is_contiguous() && bias.numel() == heads * 16 * 16,
           "bias must be (heads, 16, 16) bf16 contiguous");
  auto out = torch::empty({Bn, H, W, C}, qkv.options());
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_window_attn4(qkv.data_ptr(), out.data_ptr(),
                                 bias.data_ptr(), (int)Bn, (int)H,
                                 (int)W, (int)C, (int)heads, (int)shift,
                                 (float)scale, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

static std::vector<torch::Tensor> add_layernorm_bf16(torch::Tensor x,
                                                     torch::Tensor other,
                                                     torch::Tensor w,
                                                     torch::Tensor b,
                                                     double eps) {
  AM_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
           "x must be contiguous bf16 on GPU");
  AM_CHECK(other.is_contiguous() && other.sizes() == x.sizes() &&
               other.scalar_type() == at::kBFloat16,
           "other must match x");
  const int dim = x.size(-1);
  AM_CHECK(dim % 4 == 0 && dim <= 4096, "dim must be /4 and <= 4096");
  auto sum = torch::empty_like(x);
  auto y = torch::empty_like(x);
  const long long n_rows = x.numel() / dim;
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_add_layernorm_bf16(
      x.data_ptr(), other.data_ptr(), sum.data_ptr(), y.data_ptr(),
      w.data_ptr(), b.data_ptr(), n_rows, dim, (float)eps, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return {sum, y};
}

static torch::Tensor layernorm_bf16_fp8(torch::Tensor x, torch::Tensor w,
                                        torch::Tensor b, double eps,
                                        torch::Tensor scale,
                                        torch::Tensor amax) {
  AM_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
           "x must be contiguous bf16 on GPU");
  AM_CHECK(scale.scalar_type() == at::kFloat && amax.scalar_type() == at::kFloat,
           "scale/amax must be f32 device scalars");
  const int dim = x.size(-1);
  AM_CHECK(dim % 4 == 0 && dim <= 4096, "dim must be /4 and <= 4096");
  auto y8 = torch::empty(x.sizes(),
                         x.options().dtype(at::kFloat8_e4m3fn));
  const long long n_rows = x.numel() / dim;
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_layernorm_bf16_fp8_impl(
      x.data_ptr(), y8.data_ptr(), w.data_ptr(), b.data_ptr(), nullptr,
      nullptr, scale.data_ptr<float>(), amax.data_ptr<float>(), n_rows, dim,
      (float)eps, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return y8;
}

static std::vector<torch::Tensor> add_layernorm_bf16_fp8(
    torch::Tensor x, torch::Tensor other, torch::Tensor w, torch::Tensor b,
    double eps, torch::Tensor scale, torch::Tensor amax) {
  AM_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous(),
           "x must be contiguous bf16 on GPU");
  AM_CHECK(other.is_contiguous() && other.sizes() == x.sizes() &&
               other.scalar_type() == at::kBFloat16,
           "other must match x");
  const int dim = x.size(-1);
  AM_CHECK(dim % 4 == 0 && dim <= 4096, "dim must be /4 and <= 4096");
  auto sum = torch::empty_like(x);
  auto y8 = torch::empty(x.sizes(), x.options().dtype(at::kFloat8_e4m3fn));
  const long long n_rows = x.numel() / dim;
  auto stream = c10::hip::getCurrentHIPStream();
  audiomuse::launch_layernorm_bf16_fp8_impl(
      x.data_ptr(), y8.data_ptr(), w.data_ptr(), b.data_ptr(),
      other.data_ptr(), sum.data_ptr(), scale.data_ptr<float>(),
      amax.data_ptr<float>(), n_rows, dim, (float)eps, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return {sum, y8};
}

void register_gemm_gelu(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "AudioMuse-AMD native CDNA4 kernels";
  register_gemm_gelu(m);
  m.def("add_layernorm_bf16", &add_layernorm_bf16,
        "Fused residual add + LayerNorm: returns (x+other, LN(x+other))");
  m.def("layernorm_bf16_fp8", &layernorm_bf16_fp8,
        "LayerNorm with fused e4m3 quantize (delayed scale + amax)");
  m.def("add_layernorm_bf16_fp8", &add_layernorm_bf16_fp8,
        "Residual add + LN with fused e4m3 quantize: (sum bf16, y fp8)");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("window_attn_fp8_fwd", &window_attn_fp8_fwd,
        "fused shifted-window attention, fp8-ingest QKV (e4m3 + device "
        "dequant scale), bf16 MFMAs/out");
  m.def("window_attn_fwd", &window_attn_fwd,
        "Fused shifted-window attention (qkv BHW3C bf16, bias, heads, "
        "shift, scale) -> (B,H,W,C)");
  m.def("window_attn4_fwd", &window_attn4_fwd,
        "Fused 4x4-window attention for stage 4 (qkv BHW3C bf16, "
        "bias (heads,16,16), heads, shift, scale) -> (B,H,W,C)");
  m.def("layernorm_bf16", &layernorm_bf16,
        "Fused LayerNorm forward, bf16 in/out, fp32 stats (x, w, b, eps)");
  m.def("mel_fwd", &mel_fwd,
        "Fused STFT+mel+log spectrogram (audio, window, twiddle, rowptr, "
        "bin, w, hop, n_fft, center, log_mode)");
  m.def("ivf_scan", &ivf_scan,
        "IVF probed-cell distance scan (dtype, metric, query, qnorm, data, "
        "row_norm, probe, cell_off, cand_off, out_dist, out_row, dim_pad)");
  m.attr("gfx_arch") = "gfx950";
}
