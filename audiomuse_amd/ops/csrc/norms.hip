// Fused LayerNorm forward (bf16 in/out, fp32 stats) for gfx950.
//
// The HTSAT encoder spends ~18.5% of its step time in eager LayerNorm
// (measured: profiles/r01_bench_baseline.md). This kernel does one HBM
// read per element: each 64-lane wave owns one row, loads it vectorized
// (short4 = 4 bf16 / 8 B per lane per iteration, Guideline 13), reduces
// mean/var with wave shuffles, and writes the normalized row.
//
// NIT (per-lane short4 iterations) is a template parameter so the value
// cache stays in registers — runtime-indexed local arrays spill to
// scratch on hipcc (cdna_hip_programming.md, common-mistake #20).

#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>

namespace audiomuse {

// fp8-output variants (template flag F8): the normalized row is written
// as OCP e4m3 with a delayed per-tensor scale (read from device memory)
// while the true amax accumulates via atomicMax for the NEXT step's
// scale — this removes the standalone quantize pass that made unfused
// fp8 serving slower than bf16 (profiles/r01_final_profile.md).
__device__ __forceinline__ void am_atomic_fmax(float* addr, float v) {
  // non-negative floats compare correctly as ints
  atomicMax(reinterpret_cast<int*>(addr), __float_as_int(v));
}

// 4 floats -> 4 OCP e4m3 bytes via the gfx950 packed-convert
// instruction (the hip_fp8.h helper is software emulation)
__device__ __forceinline__ unsigned int am_pack_fp8x4(float f0, float f1,
                                                      float f2, float f3) {
  int packed = 0;
  packed = __builtin_amdgcn_cvt_pk_fp8_f32(f0, f1, packed, false);
  packed = __builtin_amdgcn_cvt_pk_fp8_f32(f2, f3, packed, true);
  return (unsigned int)packed;
}

__device__ __forceinline__ float wrsum(float v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return __shfl(v, 0, 64);
}

__device__ __forceinline__ float hrsum(float v) {
  // reduce within each 32-lane half (2 rows share one wave)
  for (int off = 16; off > 0; off >>= 1) v += __shfl_xor(v, off, 32);
  return v;
}

// dim <= 128: two rows per wave (lanes 0-31 / 32-63), full-wave utilization
// (measured: the one-row variant at dim 128 runs 80% VALUBusy on half-idle
// waves — profiles/r01_bench_clap.md).
// ADD variant fuses the preceding residual add: in2 != nullptr adds it to
// x, writes the sum to `sum`, and normalizes the sum (saves one full
// tensor read between the eager add and the LN).
template <bool ADD, bool F8 = false>
__global__ __launch_bounds__(256) void layernorm_bf16_half_kernel(
    const __hip_bfloat16* __restrict__ x, __hip_bfloat16* __restrict__ y,
    const __hip_bfloat16* __restrict__ w, const __hip_bfloat16* __restrict__ b,
    const __hip_bfloat16* __restrict__ in2, __hip_bfloat16* __restrict__ sum_out,
    long long n_rows, int dim, float eps,
    const float* __restrict__ q_scale = nullptr,
    float* __restrict__ q_amax = nullptr,
    unsigned char* __restrict__ y8 = nullptr) {
  const int sl = threadIdx.x & 31;          // lane within the half
  const long long row = (long long)blockIdx.x * 8 + (threadIdx.x >> 5);
  if (row >= n_rows) return;

  const __hip_bfloat16* xr = x + row * dim;
  __hip_bfloat16* yr = y + row * dim;
  float vals[4] = {0.f, 0.f, 0.f, 0.f};
  float sum = 0.0f;
  const int i = sl * 4;
  if (i < dim) {
    const short4 p = *reinterpret_cast<const short4*>(xr + i);
    const __hip_bfloat16* pb = reinterpret_cast<const __hip_bfloat16*>(&p);
    short4 q;
    const __hip_bfloat16* qb = reinterpret_cast<const __hip_bfloat16*>(&q);
    if (ADD) q = *reinterpret_cast<const short4*>(in2 + row * dim + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      vals[j] = __bfloat162float(pb[j]);
      if (ADD) vals[j] += __bfloat162float(qb[j]);
      sum += vals[j];
    }
    if (ADD) {
      short4 so;
      __hip_bfloat16* sb = reinterpret_cast<__hip_bfloat16*>(&so);
#pragma unroll
      for (int j = 0; j < 4; ++j) sb[j] = __float2bfloat16(vals[j]);
      *reinterpret_cast<short4*>(sum_out + row * dim + i) = so;
    }
  }
  const float mean = hrsum(sum) / dim;
  float var = 0.0f;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const float d = i < dim ? vals[j] - mean : 0.0f;
    var += d * d;
  }
  const float rstd = rsqrtf(hrsum(var) / dim + eps);
  if (i < dim) {
    const short4 pw = *reinterpret_cast<const short4*>(w + i);
    const short4 pbv = *reinterpret_cast<const short4*>(b + i);
    const __hip_bfloat16* wb = reinterpret_cast<const __hip_bfloat16*>(&pw);
    const __hip_bfloat16* bb = reinterpret_cast<const __hip_bfloat16*>(&pbv);
    if (F8) {
      const float inv = 1.0f / q_scale[0];
      float amax = 0.0f;
      float fq[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float f = ((vals[j] - mean) * rstd) * __bfloat162float(wb[j]) +
                        __bfloat162float(bb[j]);
        amax = fmaxf(amax, fabsf(f));
        fq[j] = __builtin_amdgcn_fmed3f(f * inv, 448.0f, -448.0f);
      }
      *reinterpret_cast<unsigned int*>(y8 + row * dim + i) =
          am_pack_fp8x4(fq[0], fq[1], fq[2], fq[3]);
      // amax is SAMPLED (1/64 of blocks) into a 256-slot vector: a full
      // per-wave atomic on one address serializes; the delayed scale
      // only needs a statistical amax and the e4m3 cast saturates
      if ((blockIdx.x & 63) == 0) {
        for (int off = 16; off > 0; off >>= 1)
          amax = fmaxf(amax, __shfl_xor(amax, off, 32));
        if (sl == 0) am_atomic_fmax(q_amax + (blockIdx.x & 255), amax);
      }
    } else {
      short4 out;
      __hip_bfloat16* ob = reinterpret_cast<__hip_bfloat16*>(&out);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        ob[j] = __float2bfloat16(((vals[j] - mean) * rstd) * __bfloat162float(wb[j]) + __bfloat162float(bb[j]));
      *reinterpret_cast<short4*>(yr + i) = out;
    }
  }
}

// one wave per row; block = 256 threads = 4 rows
template <int NIT, bool ADD, bool F8 = false>
__global__ __launch_bounds__(256) void layernorm_bf16_kernel(
    const __hip_bfloat16* __restrict__ x, __hip_bfloat16* __restrict__ y,
    const __hip_bfloat16* __restrict__ w, const __hip_bfloat16* __restrict__ b,
    const __hip_bfloat16* __restrict__ in2, __hip_bfloat16* __restrict__ sum_out,
    long long n_rows, int dim, float eps,
    const float* __restrict__ q_scale = nullptr,
    float* __restrict__ q_amax = nullptr,
    unsigned char* __restrict__ y8 = nullptr) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long long row = (long long)blockIdx.x * 4 + wave;
  if (row >= n_rows) return;

  const __hip_bfloat16* xr = x + row * dim;
  __hip_bfloat16* yr = y + row * dim;

  float vals[NIT * 4];
  float sum = 0.0f;
#pragma unroll
  for (int t = 0; t < NIT; ++t) {
    const int i = lane * 4 + t * 256;
    if (i < dim) {
      const short4 p = *reinterpret_cast<const short4*>(xr + i);
      const __hip_bfloat16* pb = reinterpret_cast<const __hip_bfloat16*>(&p);
      short4 q;
      const __hip_bfloat16* qb = reinterpret_cast<const __hip_bfloat16*>(&q);
      if (ADD) q = *reinterpret_cast<const short4*>(in2 + row * dim + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float f = __bfloat162float(pb[j]);
        if (ADD) f += __bfloat162float(qb[j]);
        vals[t * 4 + j] = f;
        sum += f;
      }
      if (ADD) {
        short4 so;
        __hip_bfloat16* sb = reinterpret_cast<__hip_bfloat16*>(&so);
#pragma unroll
        for (int j = 0; j < 4; ++j) sb[j] = __float2bfloat16(vals[t * 4 + j]);
        *reinterpret_cast<short4*>(sum_out + row * dim + i) = so;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) vals[t * 4 + j] = 0.0f;
    }
  }
  const float mean = wrsum(sum) / dim;
  float var = 0.0f;
#pragma unroll
  for (int t = 0; t < NIT; ++t) {
    const int i = lane * 4 + t * 256;
    if (i < dim) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float d = vals[t * 4 + j] - mean;
        var += d * d;
      }
    }
  }
  const float rstd = rsqrtf(wrsum(var) / dim + eps);

  const float inv = F8 ? 1.0f / q_scale[0] : 0.0f;
  float amax = 0.0f;
#pragma unroll
  for (int t = 0; t < NIT; ++t) {
    const int i = lane * 4 + t * 256;
    if (i >= dim) continue;
    const short4 pw = *reinterpret_cast<const short4*>(w + i);
    const short4 pbv = *reinterpret_cast<const short4*>(b + i);
    const __hip_bfloat16* wb = reinterpret_cast<const __hip_bfloat16*>(&pw);
    const __hip_bfloat16* bb = reinterpret_cast<const __hip_bfloat16*>(&pbv);
    if (F8) {
      float fq[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float f = (vals[t * 4 + j] - mean) * rstd *
                        __bfloat162float(wb[j]) + __bfloat162float(bb[j]);
        amax = fmaxf(amax, fabsf(f));
        fq[j] = __builtin_amdgcn_fmed3f(f * inv, 448.0f, -448.0f);
      }
      *reinterpret_cast<unsigned int*>(y8 + row * dim + i) =
          am_pack_fp8x4(fq[0], fq[1], fq[2], fq[3]);
    } else {
      short4 out;
      __hip_bfloat16* ob = reinterpret_cast<__hip_bfloat16*>(&out);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float f = (vals[t * 4 + j] - mean) * rstd;
        ob[j] = __float2bfloat16(f * __bfloat162float(wb[j]) + __bfloat162float(bb[j]));
      }
      *reinterpret_cast<short4*>(yr + i) = out;
    }
  }
  if (F8 && (blockIdx.x & 63) == 0) {
    for (int off = 32; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, 64));
    if (lane == 0) am_atomic_fmax(q_amax + (blockIdx.x & 255), amax);
  }
}

void launch_layernorm_bf16_impl(const void* x, void* y, const void* w,
                                const void* b, const void* in2, void* sum_out,
                                long long n_rows, int dim, float eps,
                                hipStream_t stream) {
  const bool add = in2 != nullptr;
  if (dim <= 128) {
    const long long blocks2 = (n_rows + 7) / 8;
    if (add)
      hipLaunchKernelGGL((layernorm_bf16_half_kernel<true>),
                         dim3((unsigned)blocks2), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x, (__hip_bfloat16*)y,
                         (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                         (const __hip_bfloat16*)in2, (__hip_bfloat16*)sum_out,
                         n_rows, dim, eps);
    else
      hipLaunchKernelGGL((layernorm_bf16_half_kernel<false>),
                         dim3((unsigned)blocks2), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x, (__hip_bfloat16*)y,
                         (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                         nullptr, nullptr, n_rows, dim, eps);
    return;
  }
  const long long blocks = (n_rows + 3) / 4;
  const dim3 grid((unsigned)blocks);
  const dim3 block(256);
#define AM_LN_CASE(NIT, A)                                                     \
  hipLaunchKernelGGL((layernorm_bf16_kernel<NIT, A>), grid, block, 0, stream,  \
                     (const __hip_bfloat16*)x, (__hip_bfloat16*)y,             \
                     (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,       \
                     (const __hip_bfloat16*)in2, (__hip_bfloat16*)sum_out,     \
                     n_rows, dim, eps)
#define AM_LN_DISPATCH(NIT)                                                    \
  do {                                                                         \
    if (add) AM_LN_CASE(NIT, true);                                            \
    else AM_LN_CASE(NIT, false);                                               \
  } while (0)
  const int nit = (dim + 255) / 256;
  if (nit <= 1) AM_LN_DISPATCH(1);
  else if (nit <= 2) AM_LN_DISPATCH(2);
  else if (nit <= 4) AM_LN_DISPATCH(4);
  else if (nit <= 8) AM_LN_DISPATCH(8);
  else AM_LN_DISPATCH(16);
#undef AM_LN_DISPATCH
#undef AM_LN_CASE
}

void launch_layernorm_bf16_fp8_impl(const void* x, void* y8, const void* w,
                                    const void* b, const void* in2,
                                    void* sum_out, const float* q_scale,
                                    float* q_amax, long long n_rows, int dim,
                                    float eps, hipStream_t stream) {
  const bool add = in2 != nullptr;
  if (dim <= 128) {
    const long long blocks2 = (n_rows + 7) / 8;
    if (add)
      hipLaunchKernelGGL((layernorm_bf16_half_kernel<true, true>),
                         dim3((unsigned)blocks2), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x, nullptr,
                         (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                         (const __hip_bfloat16*)in2, (__hip_bfloat16*)sum_out,
                         n_rows, dim, eps, q_scale, q_amax,
                         (unsigned char*)y8);
    else
      hipLaunchKernelGGL((layernorm_bf16_half_kernel<false, true>),
                         dim3((unsigned)blocks2), dim3(256), 0, stream,
                         (const __hip_bfloat16*)x, nullptr,
                         (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                         nullptr, nullptr, n_rows, dim, eps, q_scale, q_amax,
                         (unsigned char*)y8);
    return;
  }
  const long long blocks = (n_rows + 3) / 4;
  const dim3 grid((unsigned)blocks);
  const dim3 block(256);
#define AM_LN8_CASE(NIT, A)                                                    \
  hipLaunchKernelGGL((layernorm_bf16_kernel<NIT, A, true>), grid, block, 0,    \
                     stream, (const __hip_bfloat16*)x, nullptr,                \
                     (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,       \
                     (const __hip_bfloat16*)in2, (__hip_bfloat16*)sum_out,     \
                     n_rows, dim, eps, q_scale, q_amax, (unsigned char*)y8)
#define AM_LN8_DISPATCH(NIT)                                                   \
  do {                                                                         \
    if (add) AM_LN8_CASE(NIT, true);                                           \
    else AM_LN8_CASE(NIT, false);                                              \
  } while (0)
  const int nit = (dim + 255) / 256;
  if (nit <= 1) AM_LN8_DISPATCH(1);
  else if (nit <= 2) AM_LN8_DISPATCH(2);
  else if (nit <= 4) AM_LN8_DISPATCH(4);
  else if (nit <= 8) AM_LN8_DISPATCH(8);
  else AM_LN8_DISPATCH(16);
#undef AM_LN8_DISPATCH
#undef AM_LN8_CASE
}

void launch_layernorm_bf16(const void* x, void* y, const void* w,
                           const void* b, long long n_rows, int dim, float eps,
                           hipStream_t stream) {
  launch_layernorm_bf16_impl(x, y, w, b, nullptr, nullptr, n_rows, dim, eps,
                             stream);
}

void launch_add_layernorm_bf16(const void* x, const void* in2, void* sum_out,
                               void* y, const void* w, const void* b,
                               long long n_rows, int dim, float eps,
                               hipStream_t stream) {
  launch_layernorm_bf16_impl(x, y, w, b, in2, sum_out, n_rows, dim, eps,
                             stream);
}

}  // namespace audiomuse
