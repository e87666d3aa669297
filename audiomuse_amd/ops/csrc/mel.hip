// Fused STFT -> power -> mel -> log spectrogram kernel for gfx950 (CDNA4).
//
// Replaces the reference's librosa.feature.melspectrogram call sites
// (/root/reference/tasks/clap_analyzer.py:394-430 CLAP shape 48k/2048/480/128;
//  /root/reference/tasks/analysis/song.py:240-256 MusiCNN 16k/512/256/96)
// with one kernel launch per batch: each workgroup computes TWO frames'
// windowed 2^k-point FFT in LDS via the real-pair trick (frames packed as
// re/im of one complex FFT, split by conjugate symmetry), the power
// spectra, the sparse (CSR) slaney mel projection, and the log
// compression, writing only the (n_mels) outputs per frame to HBM.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
// - block = 256 threads (4 waves); LDS = NFFT float2 + 2*(NFFT/2+1) float
//   (24.4 KiB at NFFT=2048) -> LDS allows 6 blocks/CU (wave-capacity 8).
// - grid = ceil(n_frames/2) x B: B=256 CLAP segments -> 128k workgroups.
// - fp32 throughout: the front-end feeds catalogue identity (simhash);
//   numeric fidelity vs the librosa reference matters.
// - optional fused int16 round-trip on load (clap_analyzer.py:453-455).

#include <hip/hip_runtime.h>

namespace audiomuse {

template <int NFFT, int LOG2N>
__global__ __launch_bounds__(256) void mel_fwd_kernel(
    const float* __restrict__ audio,    // (B, T)
    float* __restrict__ out,            // (B, n_mels, n_frames)
    const float* __restrict__ window,   // (NFFT) periodic hann
    const float2* __restrict__ twiddle, // (NFFT/2) {cos, -sin}(2*pi*j/NFFT)
    const int* __restrict__ mel_rowptr, // (n_mels+1) CSR over mel bins
    const int* __restrict__ mel_bin,    // (nnz) fft-bin indices
    const float* __restrict__ mel_w,    // (nnz) filter weights
    int T, int n_frames, int hop, int n_mels, int center, int log_mode,
    int quant16) {
  // zbuf is XOR-swizzled: the bit-reversal scatter otherwise lands all 64
  // lanes on one bank pair (measured 5.7M SQ_LDS_BANK_CONFLICT/dispatch) —
  // folding bits 5-9 into the bank bits spreads lanes across all banks.
  __shared__ float2 zbuf[NFFT];
  __shared__ float pw[2][NFFT / 2 + 1];
#define AM_ZS(i) ((i) ^ (((i) >> 5) & 31))

  const int f0 = blockIdx.x * 2;       // this block's frame pair
  const int b = blockIdx.y;
  const int tid = threadIdx.x;
  if (f0 >= n_frames) return;
  const bool has_f1 = (f0 + 1) < n_frames;

  const float* src = audio + (long long)b * T;
  const int start0 = f0 * hop - (center ? NFFT / 2 : 0);

  // Phase 1: load both frames (reflect pad + optional int16 round-trip +
  // window), packed re/im, bit-reverse scatter into LDS.
  for (int i = tid; i < NFFT; i += blockDim.x) {
    float v[2];
#pragma unroll
    for (int fr = 0; fr < 2; ++fr) {
      int g = start0 + fr * hop + i;
      if (fr == 1 && !has_f1) {
        v[1] = 0.0f;
        continue;
      }
      if (g < 0) g = -g;                  // librosa reflect (no edge repeat)
      if (g >= T) g = 2 * (T - 1) - g;
      g = max(0, min(T - 1, g));          // safety for tiny T
      float x = src[g];
      if (quant16) {
        x = fminf(1.0f, fmaxf(-1.0f, x));
        x = (float)(int)(x * 32767.0f) / 32767.0f;  // numpy int16 trunc
      }
      v[fr] = x * window[i];
    }
    const int rev = __brev((unsigned)i) >> (32 - LOG2N);
    zbuf[AM_ZS(rev)] = make_float2(v[0], v[1]);
  }
  __syncthreads();

  // Phase 2: DIT FFT over bit-reversed data, THREE radix-2 stages merged
  // into one radix-8 pass (one LDS round-trip + one barrier per triple —
  // measured: the stage loop dominated the kernel's VALU+LDS time), with
  // radix-4 / radix-2 leftovers for LOG2N % 3.
  int s = 1;
  for (; s + 2 <= LOG2N; s += 3) {
    const int h = 1 << (s - 1);          // stage-s half
    for (int q = tid; q < NFFT / 8; q += blockDim.x) {
      const int grp = q >> (s - 1);
      const int j = q & (h - 1);
      const int i0 = (grp << (s + 2)) + j;
      // twiddles: w1 = W(j, 2h), w2 = W(j, 4h), w4 = W(j, 8h)
      const float2 w1 = twiddle[j * (NFFT >> s)];
      const float2 w2 = twiddle[j * (NFFT >> (s + 1))];
      const float2 w4 = twiddle[j * (NFFT >> (s + 2))];
      float2 x[8];
#pragma unroll
      for (int m = 0; m < 8; ++m) x[m] = zbuf[AM_ZS(i0 + m * h)];
      // stage s: pairs (m, m+1), all with w1
      float2 bb[8];
#pragma unroll
      for (int m = 0; m < 8; m += 2) {
        const float tr = w1.x * x[m + 1].x - w1.y * x[m + 1].y;
        const float ti = w1.x * x[m + 1].y + w1.y * x[m + 1].x;
        bb[m] = make_float2(x[m].x + tr, x[m].y + ti);
        bb[m + 1] = make_float2(x[m].x - tr, x[m].y - ti);
      }
      // stage s+1: (g, g+2) with w2, (g+1, g+3) with -i*w2
      float2 dd[8];
      const float2 w3 = make_float2(w2.y, -w2.x);    // -i * w2
#pragma unroll
      for (int g = 0; g < 8; g += 4) {
        float tr = w2.x * bb[g + 2].x - w2.y * bb[g + 2].y;
        float ti = w2.x * bb[g + 2].y + w2.y * bb[g + 2].x;
        dd[g] = make_float2(bb[g].x + tr, bb[g].y + ti);
        dd[g + 2] = make_float2(bb[g].x - tr, bb[g].y - ti);
        tr = w3.x * bb[g + 3].x - w3.y * bb[g + 3].y;
        ti = w3.x * bb[g + 3].y + w3.y * bb[g + 3].x;
        dd[g + 1] = make_float2(bb[g + 1].x + tr, bb[g + 1].y + ti);
        dd[g + 3] = make_float2(bb[g + 1].x - tr, bb[g + 1].y - ti);
      }
      // stage s+2: (m, m+4) with w4 * e^{-i pi m / 4}
      const float R = 0.70710678118654752f;
      float2 t4[4];
      t4[0] = w4;
      t4[1] = make_float2(R * (w4.x + w4.y), R * (w4.y - w4.x));
      t4[2] = make_float2(w4.y, -w4.x);
      t4[3] = make_float2(R * (w4.y - w4.x), -R * (w4.x + w4.y));
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const float tr = t4[m].x * dd[m + 4].x - t4[m].y * dd[m + 4].y;
        const float ti = t4[m].x * dd[m + 4].y + t4[m].y * dd[m + 4].x;
        zbuf[AM_ZS(i0 + m * h)] = make_float2(dd[m].x + tr, dd[m].y + ti);
        zbuf[AM_ZS(i0 + (m + 4) * h)] =
            make_float2(dd[m].x - tr, dd[m].y - ti);
      }
    }
    __syncthreads();
  }
  for (; s + 1 <= LOG2N; s += 2) {
    const int h = 1 << (s - 1);          // stage-s half
    for (int q = tid; q < NFFT / 4; q += blockDim.x) {
      const int grp = q >> (s - 1);
      const int j = q & (h - 1);
      const int i0 = (grp << (s + 1)) + j;
      // twiddles: w1 = W(j, 2h), w2 = W(j, 4h), w3 = -i * w2
      const float2 w1 = twiddle[j * (NFFT >> s)];
      const float2 w2 = twiddle[j * (NFFT >> (s + 1))];
      float2 x0 = zbuf[AM_ZS(i0)];
      float2 x1 = zbuf[AM_ZS(i0 + h)];
      float2 x2 = zbuf[AM_ZS(i0 + 2 * h)];
      float2 x3 = zbuf[AM_ZS(i0 + 3 * h)];
      // stage s: (x0,x1) and (x2,x3), both with twiddle w1
      float tr = w1.x * x1.x - w1.y * x1.y;
      float ti = w1.x * x1.y + w1.y * x1.x;
      const float2 a0 = make_float2(x0.x + tr, x0.y + ti);
      const float2 a1 = make_float2(x0.x - tr, x0.y - ti);
      tr = w1.x * x3.x - w1.y * x3.y;
      ti = w1.x * x3.y + w1.y * x3.x;
      const float2 a2 = make_float2(x2.x + tr, x2.y + ti);
      const float2 a3 = make_float2(x2.x - tr, x2.y - ti);
      // stage s+1: (a0,a2) with w2; (a1,a3) with w3 = -i*w2
      tr = w2.x * a2.x - w2.y * a2.y;
      ti = w2.x * a2.y + w2.y * a2.x;
      zbuf[AM_ZS(i0)] = make_float2(a0.x + tr, a0.y + ti);
      zbuf[AM_ZS(i0 + 2 * h)] = make_float2(a0.x - tr, a0.y - ti);
      const float w3x = w2.y, w3y = -w2.x;   // -i * w2
      tr = w3x * a3.x - w3y * a3.y;
      ti = w3x * a3.y + w3y * a3.x;
      zbuf[AM_ZS(i0 + h)] = make_float2(a1.x + tr, a1.y + ti);
      zbuf[AM_ZS(i0 + 3 * h)] = make_float2(a1.x - tr, a1.y - ti);
    }
    __syncthreads();
  }
  // leftover radix-2 stage when LOG2N is odd
  for (; s <= LOG2N; ++s) {
    const int half = 1 << (s - 1);
    const int tw_step = NFFT >> s;
    for (int bf = tid; bf < NFFT / 2; bf += blockDim.x) {
      const int grp = bf >> (s - 1);
      const int j = bf & (half - 1);
      const int i0 = (grp << s) + j;
      const int i1 = i0 + half;
      const float2 w = twiddle[j * tw_step];
      const float2 a = zbuf[AM_ZS(i0)];
      const float2 c = zbuf[AM_ZS(i1)];
      const float tr = w.x * c.x - w.y * c.y;
      const float ti = w.x * c.y + w.y * c.x;
      zbuf[AM_ZS(i0)] = make_float2(a.x + tr, a.y + ti);
      zbuf[AM_ZS(i1)] = make_float2(a.x - tr, a.y - ti);
    }
    __syncthreads();
  }

  // Phase 3: conjugate-split the packed pair and take power spectra.
  // frame0[k] = (Z[k] + conj(Z[N-k])) / 2 ; frame1[k] = (Z[k] - conj(Z[N-k])) / 2i
  for (int k = tid; k <= NFFT / 2; k += blockDim.x) {
    const int nk = (NFFT - k) & (NFFT - 1);
    const float2 zk = zbuf[AM_ZS(k)];
    const float2 zn = zbuf[AM_ZS(nk)];
    const float ar = 0.5f * (zk.x + zn.x);
    const float ai = 0.5f * (zk.y - zn.y);
    const float br = 0.5f * (zk.y + zn.y);
    const float bi = 0.5f * (zn.x - zk.x);
    pw[0][k] = ar * ar + ai * ai;
    pw[1][k] = br * br + bi * bi;
  }
  __syncthreads();

  // Phase 4: sparse mel projection + log for both frames, direct to HBM.
  for (int m = tid; m < 2 * n_mels; m += blockDim.x) {
    const int fr = m >= n_mels ? 1 : 0;
    if (fr == 1 && !has_f1) continue;
    const int mm = m - fr * n_mels;
    float acc = 0.0f;
    const int p0 = mel_rowptr[mm], p1 = mel_rowptr[mm + 1];
    for (int p = p0; p < p1; ++p) acc += pw[fr][mel_bin[p]] * mel_w[p];
    float y;
    if (log_mode == 0) {                       // librosa power_to_db, ref=1
      y = 10.0f * log10f(fmaxf(acc, 1e-10f));
    } else if (log_mode == 1) {                // musicnn log10(1+10000 x)
      y = log10f(1.0f + 10000.0f * fmaxf(acc, 0.0f));
    } else {                                   // raw power mel
      y = acc;
    }
    out[((long long)b * n_mels + mm) * n_frames + (f0 + fr)] = y;
  }
}

void launch_mel_fwd(const float* audio, float* out, const float* window,
                    const float2* twiddle, const int* mel_rowptr,
                    const int* mel_bin, const float* mel_w, int B, int T,
                    int n_frames, int hop, int n_mels, int n_fft, int center,
                    int log_mode, int quant16, hipStream_t stream) {
  dim3 grid((n_frames + 1) / 2, B);
  dim3 block(256);
#define AM_MEL_CASE(N, L)                                                     \
  case N:                                                                     \
    hipLaunchKernelGGL((mel_fwd_kernel<N, L>), grid, block, 0, stream, audio, \
                       out, window, twiddle, mel_rowptr, mel_bin, mel_w, T,   \
                       n_frames, hop, n_mels, center, log_mode, quant16);     \
    break;
  switch (n_fft) {
    AM_MEL_CASE(256, 8)
    AM_MEL_CASE(512, 9)
    AM_MEL_CASE(1024, 10)
    AM_MEL_CASE(2048, 11)
    AM_MEL_CASE(4096, 12)
    default:
      break;  // validated on the Python side
  }
#undef AM_MEL_CASE
}

}  // namespace audiomuse
