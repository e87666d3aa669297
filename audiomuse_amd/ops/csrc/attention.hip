// Fused shifted-window attention for the HTSAT encoder (gfx950, CDNA4).
//
// Replaces, in one launch: roll(-shift) -> window_partition -> per-head
// QK^T -> +rel_bias -> +shift mask -> softmax -> PV -> window_reverse ->
// roll(+shift)  (the eager chain measured at ~25% of encoder step time:
// profiles/r01_bench_clap.md).
//
// Geometry (fixed by the model design, models/htsat.py):
//   window 8x8 = 64 tokens  == one 64-lane wavefront's MFMA row space
//   head_dim 32             == one K-step of mfma_f32_16x16x32_bf16
// Each workgroup = 4 waves handles one window; wave w computes heads
// w, w+4, ... . S = QK^T is a 4x4 grid of 16x16 MFMA tiles (K=32, one
// instruction each); softmax runs in registers (rows live across 16
// lanes -> 4-step shfl_xor reduction); P is staged through a padded LDS
// tile; PV is a 4x2 grid of 16x16 tiles with K=64 (two instructions).
// Q and K fragments load straight from global memory (16 B per lane,
// L2-resident across heads); V is staged transposed in LDS.
//
// The shifted-window mask is computed inline from wrap bits: after a
// roll by -shift, two tokens of a window may attend iff both wrapped or
// neither wrapped in each axis (equivalent to the reference's 9-region
// mask construction, models/htsat.py:_shift_mask).
//
// Fragment layouts (A/B: row/col = lane&15, k = 8*(lane>>4)+j;
// C/D: col = lane&15, row = (lane>>4)*4 + reg) are verified on hardware
// by tests/test_attention.py::test_mfma_probe_layout via the mfma_probe
// binding below.

#include <hip/hip_runtime.h>

namespace audiomuse {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float fmax4(const f32x4 v) {
  return fmaxf(fmaxf(v[0], v[1]), fmaxf(v[2], v[3]));
}

// ---------------------------------------------------------------------------
// Layout probe: D = A(16x32) @ B(32x16) with the layout above; the GPU
// test compares against torch.matmul to pin the mapping (guide G9:
// asymmetric random inputs, transpose-detecting).
// ---------------------------------------------------------------------------
__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int r = lane & 15;
    const int k = 8 * (lane >> 4) + j;
    a[j] = *(const __bf16*)(A + r * 32 + k);   // A row-major (16,32)
    b[j] = *(const __bf16*)(B + k * 16 + r);   // B row-major (32,16), col=r
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    const int col = lane & 15;
    D[row * 16 + col] = c[reg];
  }
}

void launch_mfma_probe(const void* A, const void* B, float* D,
                       hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const __bf16*)A, (const __bf16*)B, D);
}

// ---------------------------------------------------------------------------
// Fused window attention
// ---------------------------------------------------------------------------

// per-wave LDS: VT[32][64] + a HALF P tile [64][32] bf16, XOR-swizzled
// instead of padded (swizzle: element col ^ (f(row)<<3) keeps 16 B
// reads aligned and spreads the 16-lane column reads across banks).
// 8 KiB/wave: r2 PMC showed the old 12 KiB (full 64x64 P) capped the
// CU at 13 waves -> 30.7% occupancy on a latency-bound kernel. The PV
// K loop now runs in two 32-column halves, the second half's softmax
// rows parked in registers while the first half's MFMAs drain.
constexpr int ATTN_WAVES = 1;
constexpr int WAVE_LDS_HALF = 32 * 64 + 64 * 32;  // bf16 elems (VT + P half)
#define AM_SWZ(row, col) (((row) << 6) + ((col) ^ (((row) & 7) << 3)))
// 32-wide swizzle for the P half-tile (8-element vectors stay aligned)
#define AM_SWZ32(row, col) (((row) << 5) + ((col) ^ (((row) & 3) << 3)))

__global__ __launch_bounds__(64 * ATTN_WAVES, 4) void window_attn_kernel(
    const __bf16* __restrict__ qkv,  // (B, H, W, 3C)
    __bf16* __restrict__ out,        // (B, H, W, C)
    const __bf16* __restrict__ bias,         // (heads, 64, 64) bf16:
    // halves the per-wave 16 KiB L2 bias stream (r2: latency-bound)
    int Bn, int H, int W, int C, int heads, int shift, float scale) {
  extern __shared__ __bf16 lds[];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const int nWw = W >> 3;
  const int nWh = H >> 3;
  const int win = blockIdx.x;
  const int b = win / (nWh * nWw);
  const int wrem = win - b * (nWh * nWw);
  const int wh = wrem / nWw;
  const int ww = wrem - wh * nWw;

  __bf16* VT = lds + wave * WAVE_LDS_HALF;
  __bf16* P = VT + 32 * 64;

  // token t (0..63) -> source coords + wrap bits (shifted windows)
  auto src_of = [&](int t, int& si, int& sj, int& wrap) {
    const int ri = t >> 3, ci = t & 7;
    int gi = wh * 8 + ri + shift;
    int gj = ww * 8 + ci + shift;
    const int wr = gi >= H;
    const int wc = gj >= W;
    si = wr ? gi - H : gi;
    sj = wc ? gj - W : gj;
    wrap = (wr << 1) | wc;
  };

  // wrap bits for every token, in registers (same for all heads)
  int my_si, my_sj, my_wrap;
  src_of(lane, my_si, my_sj, my_wrap);
  const long long my_base = (((long long)b * H + my_si) * W + my_sj) * 3 * C;

  // wrap bits of all 64 tokens as a packed pair of 64-bit masks via ballot
  const unsigned long long wrap_r_mask = __ballot(my_wrap & 2);
  const unsigned long long wrap_c_mask = __ballot(my_wrap & 1);

  // one head per wave; heads ride gridDim.y so late stages (few windows,
  // many heads) still fill all 256 CUs
  const int h = blockIdx.y * ATTN_WAVES + wave;
  if (h < heads) {
    // ---- Q (A-frags) and K (B-frags) straight from global, issued
    // FIRST so the S MFMAs can start while the V staging drains ----
    bf16x8 qf[4], kf[4];
    const int kk = 8 * (lane >> 4);   // k-offset of this lane's fragment
#pragma unroll
    for (int tr = 0; tr < 4; ++tr) {
      const int t = tr * 16 + (lane & 15);
      int si, sj, wr_;
      src_of(t, si, sj, wr_);
      const long long base = (((long long)b * H + si) * W + sj) * 3 * C + h * 32;
      qf[tr] = *(const bf16x8*)(qkv + base + kk);          // Q slice
      kf[tr] = *(const bf16x8*)(qkv + base + C + kk);      // K slice
    }

    // ---- stage V transposed: VT[d][t] = V[t][d] ----
    {
      const __bf16* vptr = qkv + my_base + 2 * C + h * 32;
      bf16x8 vv[4];
#pragma unroll
      for (int g = 0; g < 4; ++g) vv[g] = *(const bf16x8*)(vptr + g * 8);
#pragma unroll
      for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          VT[AM_SWZ(g * 8 + j, lane)] = (__bf16)vv[g][j];
    }

    // ---- S = QK^T: 4x4 tiles of 16x16, one MFMA each ----
    // Bias loads for the first row-tile are issued BEFORE the MFMAs so
    // their ~L2 latency hides under the matrix work; each row-tile then
    // prefetches the next tile's bias before processing its own
    // (double-buffered software pipeline).
    const int col_in_tile = lane & 15;
    const int row_grp = (lane >> 4) * 4;
    const __bf16* bias_base = bias + (h * 64 + row_grp) * 64 + col_in_tile;
    float bv[2][4][4];  // [buf][reg][tc]
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
#pragma unroll
      for (int tc = 0; tc < 4; ++tc)
        bv[0][reg][tc] = (float)bias_base[reg * 64 + tc * 16];

    f32x4 s[4][4];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 4; ++tc) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        s[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[tr], kf[tc],
                                                            acc, 0, 0, 0);
      }

    // ---- scale + bias + shift mask, then row softmax in registers ----
    // C-frag: col = lane&15, row = (lane>>4)*4 + reg
    float rmax[4][4];  // [tr][reg]
#pragma unroll
    for (int tr = 0; tr < 4; ++tr) {
      if (tr < 3) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
#pragma unroll
          for (int tc = 0; tc < 4; ++tc)
            bv[(tr + 1) & 1][reg][tc] =
                (float)bias_base[((tr + 1) * 16 + reg) * 64 + tc * 16];
      }
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
        const int rwrap = (((wrap_r_mask >> row) & 1ull) << 1) |
                          ((wrap_c_mask >> row) & 1ull);
        float m = -1e30f;
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const int col = tc * 16 + col_in_tile;
          const int cwrap = (((wrap_r_mask >> col) & 1ull) << 1) |
                            ((wrap_c_mask >> col) & 1ull);
          float v = s[tr][tc][reg] * scale + bv[tr & 1][reg][tc];
          if (shift && rwrap != cwrap) v = -1e30f;
          s[tr][tc][reg] = v;
          m = fmaxf(m, v);
        }
        rmax[tr][reg] = m;
      }
    }
    // reduce max across the 16 lanes holding each row
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float m = rmax[tr][reg];
#pragma unroll
        for (int d = 1; d < 16; d <<= 1)
          m = fmaxf(m, __shfl_xor(m, d, 64));
        rmax[tr][reg] = m;
      }
    float rsum[4][4];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float sum = 0.f;
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const float e = __expf(s[tr][tc][reg] - rmax[tr][reg]);
          s[tr][tc][reg] = e;
          sum += e;
        }
#pragma unroll
        for (int d = 1; d < 16; d <<= 1) sum += __shfl_xor(sum, d, 64);
        rsum[tr][reg] = sum;
      }

    // ---- write P = softmax(S) to LDS in TWO 32-column halves ----
    // Half 0 (key tokens 0..31 = tc 0,1) goes to the 4 KiB P buffer;
    // half 1 (tc 2,3) stays normalized in registers (s[][2..3]) and is
    // written after the first PV K-step drains — halving LDS per wave
    // was worth more than avoiding one extra single-wave barrier.
    __syncthreads();  // VT writes (and previous round's P reads) settled
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
        const float inv = __frcp_rn(rsum[tr][reg] + 1e-20f);
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const float pv = s[tr][tc][reg] * inv;
          if (tc < 2)
            P[AM_SWZ32(row, tc * 16 + col_in_tile)] = (__bf16)pv;
          else
            s[tr][tc][reg] = pv;  // parked for half 1
        }
      }
    __syncthreads();

    // ---- O = P @ V, K-step 0 (tokens 0..31) ----
    f32x4 o[4][2];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 2; ++tc) {
        bf16x8 pa = *(const bf16x8*)(
            P + AM_SWZ32(tr * 16 + (lane & 15), kk));
        bf16x8 vb = *(const bf16x8*)(
            VT + AM_SWZ(tc * 16 + (lane & 15), kk));
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        o[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, acc,
                                                            0, 0, 0);
      }

    // ---- park half 1 into the P buffer, then K-step 1 (tokens 32..63) --
    __syncthreads();
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
#pragma unroll
        for (int tc = 2; tc < 4; ++tc)
          P[AM_SWZ32(row, (tc - 2) * 16 + col_in_tile)] =
              (__bf16)s[tr][tc][reg];
      }
    __syncthreads();
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 2; ++tc) {
        bf16x8 pa = *(const bf16x8*)(
            P + AM_SWZ32(tr * 16 + (lane & 15), kk));
        bf16x8 vb = *(const bf16x8*)(
            VT + AM_SWZ(tc * 16 + (lane & 15), 32 + kk));
        o[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb,
                                                            o[tr][tc],
                                                            0, 0, 0);
      }

    // ---- scatter O back to (B, H, W, C) with the inverse roll ----
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int t = tr * 16 + row_grp + reg;
        int si, sj, w_;
        src_of(t, si, sj, w_);
        __bf16* op =
            out + (((long long)b * H + si) * W + sj) * C + h * 32;
#pragma unroll
        for (int tc = 0; tc < 2; ++tc)
          op[tc * 16 + col_in_tile] = (__bf16)o[tr][tc][reg];
      }
    __syncthreads();  // P/VT reuse next round
  }
}

// ---------------------------------------------------------------------------
// Fused 4x4-window attention (stage 4: grid 32x4, heads 32, window 4).
//
// 16 tokens x head_dim 32 per (window, head) wave: S = QK^T is ONE
// 16x16 MFMA (K=32); softmax in registers; O = PV is two 16x16 tiles
// whose K (=16 tokens) rides a zero-padded K=32 MFMA. Replaces the
// eager roll/partition/SDPA/reverse chain that r1 left on stage 4
// (models/htsat.py:231-246).
// ---------------------------------------------------------------------------

constexpr int A4_LDS = 32 * 32 + 16 * 32;   // VT[32][32] + P[16][32] bf16

__global__ __launch_bounds__(64, 6) void window_attn4_kernel(
    const __bf16* __restrict__ qkv,  // (B, H, W, 3C)
    __bf16* __restrict__ out,        // (B, H, W, C)
    const __bf16* __restrict__ bias,  // (heads, 16, 16) bf16
    int Bn, int H, int W, int C, int heads, int shift, float scale) {
  extern __shared__ __bf16 lds[];
  const int lane = threadIdx.x & 63;

  const int nWw = W >> 2;
  const int nWh = H >> 2;
  const int win = blockIdx.x;
  const int b = win / (nWh * nWw);
  const int wrem = win - b * (nWh * nWw);
  const int wh = wrem / nWw;
  const int ww = wrem - wh * nWw;

  __bf16* VT = lds;            // [dim 32][token 32 padded]
  __bf16* P = VT + 32 * 32;    // [row 16][k 32 padded]

  auto src_of = [&](int t, int& si, int& sj, int& wrap) {
    const int ri = t >> 2, ci = t & 3;
    int gi = wh * 4 + ri + shift;
    int gj = ww * 4 + ci + shift;
    const int wr = gi >= H;
    const int wc = gj >= W;
    si = wr ? gi - H : gi;
    sj = wc ? gj - W : gj;
    wrap = (wr << 1) | wc;
  };

  // wrap bits of the 16 tokens (lanes 0-15 vote; mask replicated)
  int si0, sj0, wrap0;
  src_of(lane & 15, si0, sj0, wrap0);
  const unsigned long long wrap_r_mask = __ballot(wrap0 & 2) & 0xffffull;
  const unsigned long long wrap_c_mask = __ballot(wrap0 & 1) & 0xffffull;

  const int h = blockIdx.y;

  // ---- zero the K-padding once (tokens 16..31) ----
  // VT cols 16..31 and P cols 16..31 must be zero for the padded-K MFMA
  {
    const int r = lane >> 1, c0 = 16 + 8 * (lane & 1);
    if (r < 32) *(bf16x8*)(VT + r * 32 + c0) = bf16x8{};
    if (r < 16) *(bf16x8*)(P + r * 32 + c0) = bf16x8{};
  }

  // ---- Q/K fragments from global ----
  // A/B-frag: row|col = lane&15, k = 8*(lane>>4)+j
  const int kk = 8 * (lane >> 4);
  const int t16 = lane & 15;
  int si, sj, w_;
  src_of(t16, si, sj, w_);
  const long long base = (((long long)b * H + si) * W + sj) * 3 * C + h * 32;
  const bf16x8 qf = *(const bf16x8*)(qkv + base + kk);
  const bf16x8 kf = *(const bf16x8*)(qkv + base + C + kk);

  // ---- stage V transposed: VT[dim][token] (16 valid tokens) ----
  if (lane < 16) {
    const __bf16* vptr = qkv + base + 2 * C;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const bf16x8 vv = *(const bf16x8*)(vptr + g * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) VT[(g * 8 + j) * 32 + t16] = vv[j];
    }
  }

  // ---- S = QK^T (one MFMA), bias prefetched alongside ----
  const __bf16* bias_base = bias + (h * 16 + (lane >> 4) * 4) * 16 + t16;
  float bv[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) bv[reg] = (float)bias_base[reg * 16];
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  f32x4 s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf, acc, 0, 0, 0);

  // ---- scale + bias + shift mask + row softmax ----
  // C-frag: col = lane&15, row = (lane>>4)*4 + reg; the 16 lanes of a
  // row live in the SAME lane group? No: a row's 16 cols spread over
  // lanes with identical (lane>>4, reg) — reduce across lane&15.
  float ex[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    const int col = t16;
    const int rwrap = (((wrap_r_mask >> row) & 1ull) << 1) |
                      ((wrap_c_mask >> row) & 1ull);
    const int cwrap = (((wrap_r_mask >> col) & 1ull) << 1) |
                      ((wrap_c_mask >> col) & 1ull);
    float v = s[reg] * scale + bv[reg];
    if (shift && rwrap != cwrap) v = -1e30f;
    float m = v;
#pragma unroll
    for (int d = 1; d < 16; d <<= 1) m = fmaxf(m, __shfl_xor(m, d, 64));
    float e = __expf(v - m);
    float sum = e;
#pragma unroll
    for (int d = 1; d < 16; d <<= 1) sum += __shfl_xor(sum, d, 64);
    ex[reg] = e * __frcp_rn(sum + 1e-20f);
  }

  __syncthreads();   // V staged before PV reads; P pad settled
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    P[row * 32 + t16] = (__bf16)ex[reg];
  }
  __syncthreads();

  // ---- O = P @ V: 2 tiles (dims 0-15, 16-31), zero-padded K ----
  const bf16x8 pa = *(const bf16x8*)(P + t16 * 32 + kk);
#pragma unroll
  for (int tc = 0; tc < 2; ++tc) {
    const bf16x8 vb = *(const bf16x8*)(VT + (tc * 16 + t16) * 32 + kk);
    f32x4 z = {0.f, 0.f, 0.f, 0.f};
    const f32x4 o = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, z,
                                                            0, 0, 0);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int t = (lane >> 4) * 4 + reg;
      int osi, osj, ow_;
      src_of(t, osi, osj, ow_);
      out[(((long long)b * H + osi) * W + osj) * C + h * 32 + tc * 16 +
          t16] = (__bf16)o[reg];
    }
  }
}

void launch_window_attn4(const void* qkv, void* out, const void* bias,
                         int Bn, int H, int W, int C, int heads, int shift,
                         float scale, hipStream_t stream) {
  const int n_windows = Bn * (H >> 2) * (W >> 2);
  hipLaunchKernelGGL(window_attn4_kernel, dim3(n_windows, heads), dim3(64),
                     A4_LDS * sizeof(__bf16), stream, (const __bf16*)qkv,
                     (__bf16*)out, (const __bf16*)bias, Bn, H, W, C, heads,
                     shift, scale);
}

void launch_window_attn(const void* qkv, void* out, const void* bias, int Bn,
                        int H, int W, int C, int heads, int shift, float scale,
                        hipStream_t stream) {
  const int n_windows = Bn * (H >> 3) * (W >> 3);
  const int head_groups = (heads + ATTN_WAVES - 1) / ATTN_WAVES;
  const size_t lds_bytes = ATTN_WAVES * WAVE_LDS_HALF * sizeof(__bf16);
  hipLaunchKernelGGL(window_attn_kernel, dim3(n_windows, head_groups),
                     dim3(64 * ATTN_WAVES), lds_bytes,
                     stream, (const __bf16*)qkv, (__bf16*)out,
                     (const __bf16*)bias, Bn, H, W, C, heads, shift, scale);
}

// ---------------------------------------------------------------------------
// fp8-ingest variant (opt-in serving mode): QKV arrives as OCP e4m3
// straight from the projection GEMM's fp8-D epilogue, HALVING the
// global gather bytes that bound this kernel (r2 PMC: latency-bound at
// MfmaUtil 3.3% — compute is free, bytes are not). The MFMAs stay bf16:
// e4m3 -> bf16 is exact, P keeps full bf16 precision (no fp8 P
// quantization), and only the Q/K/V loads shrink. The per-tensor
// dequant scale is read from device memory (delayed scaling — no host
// sync); S folds qs^2 into the softmax scale, O folds qs once.
// ---------------------------------------------------------------------------

typedef float f32x2 __attribute__((ext_vector_type(2)));

__device__ __forceinline__ bf16x8 am_fp8x8_to_bf16(unsigned long long v) {
  const unsigned int lo = (unsigned int)v, hi = (unsigned int)(v >> 32);
  const f32x2 a = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  const f32x2 b = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  const f32x2 c = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  const f32x2 d = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
  bf16x8 r;
  r[0] = (__bf16)a[0]; r[1] = (__bf16)a[1];
  r[2] = (__bf16)b[0]; r[3] = (__bf16)b[1];
  r[4] = (__bf16)c[0]; r[5] = (__bf16)c[1];
  r[6] = (__bf16)d[0]; r[7] = (__bf16)d[1];
  return r;
}

// LDS per wave: VT bytes [d 32][t 64] (2 KiB) + P half-tile bf16 (4 KiB)
constexpr int WAVE_LDS_FP8 = 32 * 64 + (64 * 32) * 2;  // bytes

__global__ __launch_bounds__(64, 4) void window_attn_fp8_kernel(
    const unsigned char* __restrict__ qkv,  // (B, H, W, 3C) e4m3
    __bf16* __restrict__ out,               // (B, H, W, C) bf16
    const __bf16* __restrict__ bias,        // (heads, 64, 64) bf16
    const float* __restrict__ qs_ptr,       // per-tensor dequant scale
    int Bn, int H, int W, int C, int heads, int shift, float sm_scale) {
  extern __shared__ unsigned char lds8[];
  const int lane = threadIdx.x & 63;

  const float qs = *qs_ptr;
  const float eff_scale = sm_scale * qs * qs;

  const int nWw = W >> 3;
  const int nWh = H >> 3;
  const int win = blockIdx.x;
  const int b = win / (nWh * nWw);
  const int wrem = win - b * (nWh * nWw);
  const int wh = wrem / nWw;
  const int ww = wrem - wh * nWw;

  unsigned char* VTb = lds8;                       // [32][64] bytes
  __bf16* P = (__bf16*)(lds8 + 32 * 64);           // [64][32] bf16 half

  auto src_of = [&](int t, int& si, int& sj, int& wrap) {
    const int ri = t >> 3, ci = t & 7;
    int gi = wh * 8 + ri + shift;
    int gj = ww * 8 + ci + shift;
    const int wr = gi >= H;
    const int wc = gj >= W;
    si = wr ? gi - H : gi;
    sj = wc ? gj - W : gj;
    wrap = (wr << 1) | wc;
  };

  int my_si, my_sj, my_wrap;
  src_of(lane, my_si, my_sj, my_wrap);
  const long long my_base = (((long long)b * H + my_si) * W + my_sj) * 3 * C;
  const unsigned long long wrap_r_mask = __ballot(my_wrap & 2);
  const unsigned long long wrap_c_mask = __ballot(my_wrap & 1);

  const int h = blockIdx.y;
  if (h < heads) {
    // ---- Q/K fragments: 8 BYTES per frag (was 16), converted to bf16
    // in registers (exact) ----
    bf16x8 qf[4], kf[4];
    const int kk = 8 * (lane >> 4);
#pragma unroll
    for (int tr = 0; tr < 4; ++tr) {
      const int t = tr * 16 + (lane & 15);
      int si, sj, wr_;
      src_of(t, si, sj, wr_);
      const long long base =
          (((long long)b * H + si) * W + sj) * 3 * C + h * 32;
      qf[tr] = am_fp8x8_to_bf16(*(const unsigned long long*)(qkv + base + kk));
      kf[tr] = am_fp8x8_to_bf16(
          *(const unsigned long long*)(qkv + base + C + kk));
    }

    // ---- stage V transposed as raw e4m3 bytes: VT[d][t] ----
    {
      const unsigned char* vptr = qkv + my_base + 2 * C + h * 32;
      unsigned long long vv[4];
#pragma unroll
      for (int g = 0; g < 4; ++g)
        vv[g] = *(const unsigned long long*)(vptr + g * 8);
#pragma unroll
      for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          VTb[AM_SWZ(g * 8 + j, lane)] = (unsigned char)(vv[g] >> (8 * j));
    }

    // ---- S = QK^T, bias double-buffered exactly as the bf16 kernel ----
    const int col_in_tile = lane & 15;
    const int row_grp = (lane >> 4) * 4;
    const __bf16* bias_base = bias + (h * 64 + row_grp) * 64 + col_in_tile;
    float bv[2][4][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
#pragma unroll
      for (int tc = 0; tc < 4; ++tc)
        bv[0][reg][tc] = (float)bias_base[reg * 64 + tc * 16];

    f32x4 s[4][4];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 4; ++tc) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        s[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[tr], kf[tc],
                                                            acc, 0, 0, 0);
      }

    float rmax[4][4];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr) {
      if (tr < 3) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
#pragma unroll
          for (int tc = 0; tc < 4; ++tc)
            bv[(tr + 1) & 1][reg][tc] =
                (float)bias_base[((tr + 1) * 16 + reg) * 64 + tc * 16];
      }
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
        const int rwrap = (((wrap_r_mask >> row) & 1ull) << 1) |
                          ((wrap_c_mask >> row) & 1ull);
        float m = -1e30f;
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const int col = tc * 16 + col_in_tile;
          const int cwrap = (((wrap_r_mask >> col) & 1ull) << 1) |
                            ((wrap_c_mask >> col) & 1ull);
          float v = s[tr][tc][reg] * eff_scale + bv[tr & 1][reg][tc];
          if (shift && rwrap != cwrap) v = -1e30f;
          s[tr][tc][reg] = v;
          m = fmaxf(m, v);
        }
        rmax[tr][reg] = m;
      }
    }
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float m = rmax[tr][reg];
#pragma unroll
        for (int d = 1; d < 16; d <<= 1)
          m = fmaxf(m, __shfl_xor(m, d, 64));
        rmax[tr][reg] = m;
      }
    float rsum[4][4];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float sum = 0.f;
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const float e = __expf(s[tr][tc][reg] - rmax[tr][reg]);
          s[tr][tc][reg] = e;
          sum += e;
        }
#pragma unroll
        for (int d = 1; d < 16; d <<= 1) sum += __shfl_xor(sum, d, 64);
        rsum[tr][reg] = sum;
      }

    // ---- P half 0 to LDS (bf16), half 1 parked in registers ----
    __syncthreads();
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
        const float inv = __frcp_rn(rsum[tr][reg] + 1e-20f);
#pragma unroll
        for (int tc = 0; tc < 4; ++tc) {
          const float pv = s[tr][tc][reg] * inv;
          if (tc < 2)
            P[AM_SWZ32(row, tc * 16 + col_in_tile)] = (__bf16)pv;
          else
            s[tr][tc][reg] = pv;
        }
      }
    __syncthreads();

    // ---- O = P @ V, K-step 0 (V fragments convert from LDS bytes) ----
    f32x4 o[4][2];
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 2; ++tc) {
        bf16x8 pa = *(const bf16x8*)(
            P + AM_SWZ32(tr * 16 + (lane & 15), kk));
        bf16x8 vb = am_fp8x8_to_bf16(*(const unsigned long long*)(
            VTb + AM_SWZ(tc * 16 + (lane & 15), kk)));
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        o[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, acc,
                                                            0, 0, 0);
      }

    __syncthreads();
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = tr * 16 + row_grp + reg;
#pragma unroll
        for (int tc = 2; tc < 4; ++tc)
          P[AM_SWZ32(row, (tc - 2) * 16 + col_in_tile)] =
              (__bf16)s[tr][tc][reg];
      }
    __syncthreads();
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int tc = 0; tc < 2; ++tc) {
        bf16x8 pa = *(const bf16x8*)(
            P + AM_SWZ32(tr * 16 + (lane & 15), kk));
        bf16x8 vb = am_fp8x8_to_bf16(*(const unsigned long long*)(
            VTb + AM_SWZ(tc * 16 + (lane & 15), 32 + kk)));
        o[tr][tc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb,
                                                            o[tr][tc],
                                                            0, 0, 0);
      }

    // ---- scatter O (dequantized by qs) ----
#pragma unroll
    for (int tr = 0; tr < 4; ++tr)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int t = tr * 16 + row_grp + reg;
        int si, sj, w_;
        src_of(t, si, sj, w_);
        __bf16* op =
            out + (((long long)b * H + si) * W + sj) * C + h * 32;
#pragma unroll
        for (int tc = 0; tc < 2; ++tc)
          op[tc * 16 + col_in_tile] = (__bf16)(o[tr][tc][reg] * qs);
      }
    __syncthreads();
  }
}

void launch_window_attn_fp8(const void* qkv, void* out, const void* bias,
                            const void* qs_ptr, int Bn, int H, int W, int C,
                            int heads, int shift, float sm_scale,
                            hipStream_t stream) {
  const int n_windows = Bn * (H >> 3) * (W >> 3);
  hipLaunchKernelGGL(window_attn_fp8_kernel, dim3(n_windows, heads),
                     dim3(64), WAVE_LDS_FP8, stream,
                     (const unsigned char*)qkv, (__bf16*)out,
                     (const __bf16*)bias, (const float*)qs_ptr, Bn, H, W, C,
                     heads, shift, sm_scale);
}

}  // namespace audiomuse
