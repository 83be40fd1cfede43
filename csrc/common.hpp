// Shared declarations for the MI355X-native convolution engine.
//
// Built for gfx950 (CDNA4) only — no CUDA compatibility paths
// (BASELINE.json north-star: "no Triton, no CUDA-compat shims, no dual
// code paths").
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <stdexcept>
#include <string>
#include <vector>

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +   \
                               std::to_string(__LINE__));                      \
    }                                                                          \
  } while (0)

namespace pconv {

enum class DType : int { U8 = 0, F16 = 1, F32 = 2 };

inline int itemsize(DType d) { return d == DType::U8 ? 1 : d == DType::F16 ? 2 : 4; }

// Geometry of one rank's padded block buffer. All x-quantities are in
// storage ELEMENTS (pixel x * channels) unless suffixed Px (pixels).
// The owned block occupies rows [haloPx, haloPx+h) and element columns
// [haloPx*S, (haloPx+w)*S) of a (bufh, pitchE) buffer.
struct BlockGeom {
  int h = 0;        // owned rows
  int wPx = 0;      // owned width in pixels
  int haloPx = 0;   // allocated halo width in pixels (>= max exchange depth)
  int S = 1;        // channels (element stride of one pixel)
  int pitchE = 0;   // row pitch in elements
  int bufh = 0;     // buffer rows = h + 2*haloPx
  // neighbors: index meaning depends on engine mode (local block index or
  // global NCCL rank); -1 = physical edge. Order: N, S, W, E.
  int nbr[4] = {-1, -1, -1, -1};
};

enum { NB_N = 0, NB_S = 1, NB_W = 2, NB_E = 3 };

// Compute region in buffer coords: rows [py0,py1), element cols [ex0,ex1).
struct Region {
  int py0, py1, ex0, ex1;
};

// Launch the k×k convolution over a region.
//   src/dst: element pointers to the padded buffers
//   wgt: k*k fp32 filter on device
// Fast LDS-tiled path for radius<=2 and S in {1,3}; generic fallback
// otherwise.
void launch_conv(DType dt, int S, int k, const void* src, void* dst,
                 const float* wgt, const BlockGeom& g, const Region& r,
                 bool useMfma, hipStream_t stream);

// Element-indexed strip launch (conv_generic_k) for thin border/band
// regions where the 1024-element tiled kernels waste ~30x on masked
// lanes. Bit-compatible with every fast path on their eligible inputs:
// fp32 fma in row-major tap order with the same store rounding — for
// the dyadic u8 class all sums are fp32-exact, so integer RNE == this.
void launch_conv_strip(DType dt, int S, int k, const void* src, void* dst,
                       const float* wgt, const BlockGeom& g, const Region& r,
                       hipStream_t stream);

// Integer-dot u8 fast path (k=3, S=1, non-negative dyadic filter
// w = n/2^shift, n in [0,255], sum(n) <= 2^shift): exact vs the fp32
// contract. w9 = the nine integer taps row-major.
void launch_conv_u8dot(const void* src, void* dst, const unsigned* w9,
                       int shift, const BlockGeom& g, const Region& r,
                       hipStream_t stream);

// Separable integer u8 fast path (k in {3,5}, S=1, non-negative
// separable dyadic filter w = outer(b, a)/2^shift with
// 255*sum(a)*sum(b) + 2^(shift-1) <= 65535): exact vs the fp32
// contract. sep = [shift, a0..a4, b0..b4] (unused taps zero).
void launch_conv_u8sep(const void* src, void* dst, const int* sep, int k,
                       const BlockGeom& g, const Region& r,
                       hipStream_t stream);

// Fused integer-dot double iteration (u8 mid image in LDS)
void launch_conv_u8dot2(const void* src, void* dst, const unsigned* w9,
                        int shift, const BlockGeom& g, const Region& rm,
                        const Region& ro, hipStream_t stream);

// Fused double iteration: iteration A into an LDS mid image (quantized),
// iteration B to global — rm is iteration A's region, r the output
// region. Returns false if no fused template exists for (k, S).
bool launch_conv2(DType dt, int S, int k, const void* src, void* dst,
                  const float* wgt, const BlockGeom& g, const Region& rm,
                  const Region& ro, hipStream_t stream);

// Strided<->contiguous 2D device copy (halo pack/unpack):
// hipMemcpy2DAsync's SDMA path measured ~8x slower than a plain copy
// kernel for the narrow tall E/W strips (bench/exchange_cost.py).
void launch_copy2d(void* dst, size_t dpitch, const void* src, size_t spitch,
                   size_t widthB, size_t rows, hipStream_t stream);

// one v_mfma_f32_16x16x32_f16 with the conv kernel's assumed fragment
// layout: D[16][16] = A[16][32] . B[32][16] (layout verification probe)
void launch_mfma_probe(const float* A32, const float* B32, float* D,
                       hipStream_t stream);

// max |dst - src| over the owned region, accumulated into *resultDev
// (a single float, caller zeroes it first).
void launch_diff_max(DType dt, const void* src, const void* dst,
                     const BlockGeom& g, float* resultDev, hipStream_t stream);

}  // namespace pconv
