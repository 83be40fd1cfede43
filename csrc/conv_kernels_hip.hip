#include "hip/hip_runtime.h"
// CDNA4 (gfx950) k×k iterative-convolution kernels.
//
// Replaces the reference's `convolute` inner loops (SURVEY.md §2.1
// R2-R4 / §2.4 E1-E2; the reference tree was unmounted — SURVEY.md §0 —
// so the behavioral contract is BASELINE.json:5 and spec.py).
//
// Design (guided by /opt/skills/guides/cdna_hip_programming.md):
//   * 256-thread workgroups (4 waves of 64), tile = 128 output ELEMENTS
//     (x) × 32 rows; each thread computes 8 consecutive elements × 2 rows.
//   * Input staged through LDS as fp32: global loads are 16-byte
//     vectorized (Guideline 13), converted once per element (u8 →
//     v_cvt_f32_ubyte*, f16 → cvt_f32_f16), so each element's 9/25 taps
//     re-read converted data from LDS instead of re-converting.
//   * LDS rows padded to 164 dwords (160-dword window + 4) so the
//     ds_read_b128 lane groups land on distinct banks (Guideline 4; row
//     stride ≡ 36 mod 64).
//   * fp32 accumulation in row-major tap order; storage rounding per
//     spec.py (u8: round-half-even + clamp via __float2int_rn).
//   * Grayscale and packed-RGB share one template: the channel count S is
//     the x-stride of a tap in ELEMENT space (interleaved storage,
//     BASELINE.json:5), so RGB needs no de-interleave pass at all.
//   * Tiles snap to 16-element x-origins so every staging load is
//     16B-aligned regardless of the (halo-deepened) region offset.
//
// The memory-bound roofline at the headline config is ≈3.1 Tpx-it/s per
// GPU; the arithmetic here budgets ≈12-14 VALU ops per
// element so the kernel stays memory-bound.

#include "common.hpp"

#include <hip/hip_fp16.h>

namespace pconv {

namespace {

constexpr int TILE_X = 128;  // output elements per tile (x)
constexpr int TILE_Y = 32;   // output rows per tile
constexpr int NTHREADS = 256;
constexpr int WIN_X = TILE_X + 32;       // staged window elements
constexpr int LDS_PITCH = WIN_X + 4;     // +4 dwords: bank-conflict pad
constexpr int CHUNKS_PER_ROW = WIN_X / 16;

// ---- dtype traits ---------------------------------------------------------

struct U8T {
  using T = unsigned char;
  static __device__ __forceinline__ void load16(const T* p, float* f) {
    uint4 v = *reinterpret_cast<const uint4*>(p);
    const unsigned* d = reinterpret_cast<const unsigned*>(&v);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int b = 0; b < 4; ++b)
        f[4 * i + b] = (float)((d[i] >> (8 * b)) & 0xffu);
    }
  }
  static __device__ __forceinline__ unsigned q1(float v) {
    return (unsigned)__float2int_rn(fminf(fmaxf(v, 0.f), 255.f));
  }
  static __device__ __forceinline__ void store8(T* p, const float* a) {
    unsigned lo = q1(a[0]) | (q1(a[1]) << 8) | (q1(a[2]) << 16) | (q1(a[3]) << 24);
    unsigned hi = q1(a[4]) | (q1(a[5]) << 8) | (q1(a[6]) << 16) | (q1(a[7]) << 24);
    *reinterpret_cast<uint2*>(p) = make_uint2(lo, hi);
  }
  static __device__ __forceinline__ void store1(T* p, float v) { *p = (T)q1(v); }
  static __device__ __forceinline__ float toF(T v) { return (float)v; }
};

struct F16T {
  using T = __half;
  static __device__ __forceinline__ void load16(const T* p, float* f) {
    const __half2* h = reinterpret_cast<const __half2*>(p);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float2 v = __half22float2(h[i]);
      f[2 * i] = v.x;
      f[2 * i + 1] = v.y;
    }
  }
  static __device__ __forceinline__ void store8(T* p, const float* a) {
    __half2* h = reinterpret_cast<__half2*>(p);
#pragma unroll
    for (int i = 0; i < 4; ++i) h[i] = __floats2half2_rn(a[2 * i], a[2 * i + 1]);
  }
  static __device__ __forceinline__ void store1(T* p, float v) { *p = __float2half_rn(v); }
  static __device__ __forceinline__ float toF(T v) { return __half2float(v); }
};

struct F32T {
  using T = float;
  static __device__ __forceinline__ void load16(const T* p, float* f) {
    const float4* q = reinterpret_cast<const float4*>(p);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float4 v = q[i];
      f[4 * i] = v.x;
      f[4 * i + 1] = v.y;
      f[4 * i + 2] = v.z;
      f[4 * i + 3] = v.w;
    }
  }
  static __device__ __forceinline__ void store8(T* p, const float* a) {
    float4* q = reinterpret_cast<float4*>(p);
    q[0] = make_float4(a[0], a[1], a[2], a[3]);
    q[1] = make_float4(a[4], a[5], a[6], a[7]);
  }
  static __device__ __forceinline__ void store1(T* p, float v) { *p = v; }
  static __device__ __forceinline__ float toF(T v) { return v; }
};

// ---- fast LDS-tiled kernel ------------------------------------------------

// R: filter radius; S: element stride of one pixel (1 gray, 3 packed RGB).
template <typename TT, int R, int S>
__global__ __launch_bounds__(NTHREADS) void conv_tile_k(
    const typename TT::T* __restrict__ src, typename TT::T* __restrict__ dst,
    const float* __restrict__ wgt, int pitchE, int bufh, int ex0, int ex1,
    int py0, int py1, int abase, int tilesX) {
  using T = typename TT::T;
  constexpr int K = 2 * R + 1;
  constexpr int ROWS_IN = TILE_Y + 2 * R;
  // aligned window geometry: out element e reads [e-S*R, e+S*R]; with 8
  // elements per thread the aligned fp32 window is NB 4-dword blocks
  // starting A0 dwords left of the first output element.
  constexpr int A0 = ((S * R + 3) / 4) * 4;
  constexpr int NB = (A0 + 8 + S * R + 3) / 4;

  __shared__ float lds[ROWS_IN][LDS_PITCH];

  const int tid = threadIdx.x;
  const int bx = blockIdx.x % tilesX;
  const int by = blockIdx.x / tilesX;
  const int etx0 = abase + TILE_X * bx;  // tile origin (16-aligned element)
  const int gx0 = etx0 - 16;             // staged window origin
  const int ty0 = py0 + TILE_Y * by;
  const int gy0 = ty0 - R;

  // filter → registers (wave-uniform address → scalar loads)
  float w[K * K];
#pragma unroll
  for (int i = 0; i < K * K; ++i) w[i] = wgt[i];

  // stage window (ROWS_IN rows × WIN_X elements) into LDS as fp32
  for (int i = tid; i < ROWS_IN * CHUNKS_PER_ROW; i += NTHREADS) {
    const int rr = i / CHUNKS_PER_ROW;
    const int cc = i % CHUNKS_PER_ROW;
    const int gy = min(gy0 + rr, bufh - 1);
    int ga = gx0 + 16 * cc;
    ga = max(0, min(ga, pitchE - 16));
    float f[16];
    TT::load16(src + (size_t)gy * pitchE + ga, f);
    float4* out = reinterpret_cast<float4*>(&lds[rr][16 * cc]);
#pragma unroll
    for (int q = 0; q < 4; ++q)
      out[q] = make_float4(f[4 * q], f[4 * q + 1], f[4 * q + 2], f[4 * q + 3]);
  }
  __syncthreads();

  // each thread: 8 consecutive elements × 2 rows
  const int tx = tid % 16;
  const int tyy = tid / 16;
  const int ex = etx0 + 8 * tx;
  const int lx = (16 + 8 * tx) - A0;  // LDS col of the window block start

  float win[2 + 2 * R][NB * 4];
#pragma unroll
  for (int rr = 0; rr < 2 + 2 * R; ++rr) {
    const float4* row =
        reinterpret_cast<const float4*>(&lds[2 * tyy + rr][lx]);
#pragma unroll
    for (int b = 0; b < NB; ++b) {
      float4 v = row[b];
      win[rr][4 * b] = v.x;
      win[rr][4 * b + 1] = v.y;
      win[rr][4 * b + 2] = v.z;
      win[rr][4 * b + 3] = v.w;
    }
  }

  float acc[2][8];
#pragma unroll
  for (int j = 0; j < 2; ++j)
#pragma unroll
    for (int p = 0; p < 8; ++p) acc[j][p] = 0.f;

#pragma unroll
  for (int dy = 0; dy < K; ++dy)
#pragma unroll
    for (int dx = 0; dx < K; ++dx) {
      const float c = w[dy * K + dx];
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int p = 0; p < 8; ++p)
          acc[j][p] = fmaf(c, win[j + dy][A0 - S * R + p + S * dx], acc[j][p]);
    }

#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int y = ty0 + 2 * tyy + j;
    if (y >= py1) continue;
    T* out = dst + (size_t)y * pitchE + ex;
    if (ex >= ex0 && ex + 8 <= ex1) {
      TT::store8(out, acc[j]);
    } else {
#pragma unroll
      for (int p = 0; p < 8; ++p)
        if (ex + p >= ex0 && ex + p < ex1) TT::store1(out + p, acc[j][p]);
    }
  }
}

// ---- generic fallback (any odd k, any channel stride) ---------------------

template <typename TT>
__global__ __launch_bounds__(NTHREADS) void conv_generic_k(
    const typename TT::T* __restrict__ src, typename TT::T* __restrict__ dst,
    const float* __restrict__ wgt, int pitchE, int S, int R, int ex0, int ex1,
    int py0, int py1) {
  const int ew = ex1 - ex0;
  const long total = (long)(py1 - py0) * ew;
  for (long idx = (long)blockIdx.x * NTHREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * NTHREADS) {
    const int y = py0 + (int)(idx / ew);
    const int e = ex0 + (int)(idx % ew);
    const int K = 2 * R + 1;
    float acc = 0.f;
    for (int dy = 0; dy < K; ++dy)
      for (int dx = 0; dx < K; ++dx)
        acc = fmaf(wgt[dy * K + dx],
                   TT::toF(src[(size_t)(y + dy - R) * pitchE + e + S * (dx - R)]),
                   acc);
    TT::store1(dst + (size_t)y * pitchE + e, acc);
  }
}

// ---- diff-max (convergence check) ---------------------

__device__ __forceinline__ void atomicMaxFloatPos(float* addr, float v) {
  // positive floats compare like their bit patterns
  atomicMax(reinterpret_cast<unsigned*>(addr), __float_as_uint(v));
}

template <typename TT>
__global__ __launch_bounds__(NTHREADS) void diff_max_k(
    const typename TT::T* __restrict__ a, const typename TT::T* __restrict__ b,
    int pitchE, int ex0, int ex1, int py0, int py1, float* result) {
  const int ew = ex1 - ex0;
  const long total = (long)(py1 - py0) * ew;
  float m = 0.f;
  for (long idx = (long)blockIdx.x * NTHREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * NTHREADS) {
    const int y = py0 + (int)(idx / ew);
    const int e = ex0 + (int)(idx % ew);
    const size_t off = (size_t)y * pitchE + e;
    m = fmaxf(m, fabsf(TT::toF(a[off]) - TT::toF(b[off])));
  }
#pragma unroll
  for (int s = 32; s >= 1; s >>= 1) m = fmaxf(m, __shfl_down(m, s, 64));
  if ((threadIdx.x & 63) == 0 && m > 0.f) atomicMaxFloatPos(result, m);
}

// ---- dispatch -------------------------------------------------------------

template <typename TT, int R, int S>
void launch_tile(const void* src, void* dst, const float* wgt,
                 const BlockGeom& g, const Region& r, hipStream_t stream) {
  const int abase = (r.ex0 / 16) * 16;
  const int tilesX = (r.ex1 - abase + TILE_X - 1) / TILE_X;
  const int tilesY = (r.py1 - r.py0 + TILE_Y - 1) / TILE_Y;
  hipLaunchKernelGGL((conv_tile_k<TT, R, S>), dim3(tilesX * tilesY),
                     dim3(NTHREADS), 0, stream,
                     reinterpret_cast<const typename TT::T*>(src),
                     reinterpret_cast<typename TT::T*>(dst), wgt, g.pitchE,
                     g.bufh, r.ex0, r.ex1, r.py0, r.py1, abase, tilesX);
}

template <typename TT>
void launch_dtype(int S, int k, const void* src, void* dst, const float* wgt,
                  const BlockGeom& g, const Region& r, hipStream_t stream) {
  const int R = k / 2;
  if (R == 1 && S == 1) return launch_tile<TT, 1, 1>(src, dst, wgt, g, r, stream);
  if (R == 2 && S == 1) return launch_tile<TT, 2, 1>(src, dst, wgt, g, r, stream);
  if (R == 1 && S == 3) return launch_tile<TT, 1, 3>(src, dst, wgt, g, r, stream);
  if (R == 2 && S == 3) return launch_tile<TT, 2, 3>(src, dst, wgt, g, r, stream);
  // generic fallback
  const long total = (long)(r.py1 - r.py0) * (r.ex1 - r.ex0);
  int blocks = (int)std::min<long>((total + NTHREADS - 1) / NTHREADS, 2048);
  blocks = std::max(blocks, 1);
  hipLaunchKernelGGL((conv_generic_k<TT>), dim3(blocks), dim3(NTHREADS), 0,
                     stream, reinterpret_cast<const typename TT::T*>(src),
                     reinterpret_cast<typename TT::T*>(dst), wgt, g.pitchE, S,
                     R, r.ex0, r.ex1, r.py0, r.py1);
}

}  // namespace

void launch_conv(DType dt, int S, int k, const void* src, void* dst,
                 const float* wgt, const BlockGeom& g, const Region& r,
                 hipStream_t stream) {
  if (r.py1 <= r.py0 || r.ex1 <= r.ex0) return;
  switch (dt) {
    case DType::U8:
      return launch_dtype<U8T>(S, k, src, dst, wgt, g, r, stream);
    case DType::F16:
      return launch_dtype<F16T>(S, k, src, dst, wgt, g, r, stream);
    case DType::F32:
      return launch_dtype<F32T>(S, k, src, dst, wgt, g, r, stream);
  }
}

void launch_diff_max(DType dt, const void* src, const void* dst,
                     const BlockGeom& g, float* resultDev, hipStream_t stream) {
  Region r{g.haloPx, g.haloPx + g.h, g.haloPx * g.S, (g.haloPx + g.wPx) * g.S};
  const long total = (long)(r.py1 - r.py0) * (r.ex1 - r.ex0);
  int blocks = (int)std::min<long>((total + NTHREADS - 1) / NTHREADS, 2048);
  blocks = std::max(blocks, 1);
  switch (dt) {
    case DType::U8:
      hipLaunchKernelGGL((diff_max_k<U8T>), dim3(blocks), dim3(NTHREADS), 0,
                         stream, (const unsigned char*)src,
                         (const unsigned char*)dst, g.pitchE, r.ex0, r.ex1,
                         r.py0, r.py1, resultDev);
      break;
    case DType::F16:
      hipLaunchKernelGGL((diff_max_k<F16T>), dim3(blocks), dim3(NTHREADS), 0,
                         stream, (const __half*)src, (const __half*)dst,
                         g.pitchE, r.ex0, r.ex1, r.py0, r.py1, resultDev);
      break;
    case DType::F32:
      hipLaunchKernelGGL((diff_max_k<F32T>), dim3(blocks), dim3(NTHREADS), 0,
                         stream, (const float*)src, (const float*)dst, g.pitchE,
                         r.ex0, r.ex1, r.py0, r.py1, resultDev);
      break;
  }
}

}  // namespace pconv
