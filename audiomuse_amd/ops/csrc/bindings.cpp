// PyTorch bindings for the AudioMuse-AMD native kernel library (gfx950).
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

namespace audiomuse {
void launch_mel_fwd(const float* audio, float* out, const float* window,
                    const float2* twiddle, const int* mel_rowptr,
                    const int* mel_bin, const float* mel_w, int B, int T,
                    int n_frames, int hop, int n_mels, int n_fft, int center,
                    int log_mode, hipStream_t stream);
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
                             int64_t log_mode) {
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
      (int)n_mels, (int)n_fft, center ? 1 : 0, (int)log_mode, stream.stream());
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "AudioMuse-AMD native CDNA4 kernels";
  m.def("mel_fwd", &mel_fwd,
        "Fused STFT+mel+log spectrogram (audio, window, twiddle, rowptr, "
        "bin, w, hop, n_fft, center, log_mode)");
  m.attr("gfx_arch") = "gfx950";
}
