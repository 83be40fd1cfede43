// Per-GPU runtime: buffers, streams, chunked iterate/exchange loop.
//
// MI355X-native replacement for the reference's MPI main loop
// (Irecv/Isend → convolute → Waitall → border convolute →
// swap). Differences, by design (SURVEY.md §7 hard parts):
//
//  * Halo deepening: exchange depth d = r*T once per T iterations, then T
//    kernel launches over shrinking regions — amortizes RCCL latency,
//    which on xGMI dominates the KB-scale halo volume.
//  * Two-phase corner-free exchange (parallel/decomp.py docstring):
//    phase A = E/W column strips (strided → staged via pack kernels,
//    launch_copy2d — the SDMA path measured ~8x slower), phase B = N/S
//    row strips at full pitch (contiguous, sent in place).
//  * Exchange-overlap pyramid: the INTERIORS of the chunk's first m
//    iterations run on the compute stream while the exchange flies on
//    the comm stream; their border bands replay afterwards against the
//    historical buffer pairs (m = PCONV_OVERLAP_DEPTH, default 2 —
//    sized for an exchange roughly two interior-iterations long).
//  * Everything is enqueued asynchronously on two HIP streams (compute +
//    comm) with event ordering; run() performs ZERO host synchronization
//    until the end of the whole iteration loop.
//
// Communication backends:
//  * LOCAL: all blocks of the decomposition live in this process on one
//    GPU; "exchange" = device-to-device copies. This is the test mode that
//    validates pack/geometry on a 1-GPU box.
//  * RCCL: one block, neighbors are global ranks; grouped ncclSend/Recv
//    over xGMI (SURVEY.md §2.3 mapping), communicator built from a
//    Python-broadcast ncclUniqueId.

#pragma once

#include "common.hpp"

#include <rccl/rccl.h>

#include <algorithm>
#include <utility>
#include <cstdlib>
#include <cstring>

#define NCCL_CHECK(expr)                                                      \
  do {                                                                        \
    ncclResult_t _r = (expr);                                                 \
    if (_r != ncclSuccess) {                                                  \
      throw std::runtime_error(std::string("RCCL error: ") +                  \
                               ncclGetErrorString(_r) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                     \
    }                                                                         \
  } while (0)

namespace pconv {

struct DeviceBlock {
  BlockGeom g;
  void* src = nullptr;  // swapped every iteration
  void* dst = nullptr;
  // staging for phase-A column strips (E/W): send+recv, each
  // h * depth_max * S elements
  void* sendW = nullptr;
  void* sendE = nullptr;
  void* recvW = nullptr;
  void* recvE = nullptr;
  // send staging for phase-B row strips (N/S), haloPx * pitchE elements
  // each: sends must be CAPTURED before the overlap pyramid's interior
  // j>=1 overwrites the source buffer's edge strips (write-after-read
  // race, caught by the r02 fuzz campaign at overlap depth 2)
  void* sendN = nullptr;
  void* sendS = nullptr;
};

class ConvEngine {
 public:
  ConvEngine(int dtype, int k, const std::vector<float>& filter, int channels,
             bool useMfma = false, std::vector<int> dotW = {},
             std::vector<int> sepW = {})
      : dt_(static_cast<DType>(dtype)), k_(k), S_(channels), useMfma_(useMfma) {
    if (dotW.size() == 10 && dt_ == DType::U8 && k == 3 && channels == 1) {
      dotShift_ = dotW[0];
      for (int i = 0; i < 9; ++i) dotW_[i] = (unsigned)dotW[1 + i];
    }
    // separable integer path (round 2 headline): [shift, a0..a4, b0..b4]
    if (sepW.size() == 11 && dt_ == DType::U8 && (k == 3 || k == 5) &&
        channels == 1) {
      sepOn_ = true;
      for (int i = 0; i < 11; ++i) sepW_[i] = sepW[i];
    }
    if ((int)filter.size() != k * k) throw std::runtime_error("filter size != k*k");
    HIP_CHECK(hipStreamCreateWithFlags(&compute_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&comm_, hipStreamNonBlocking));
    HIP_CHECK(hipEventCreateWithFlags(&evComm_, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&evCompute_, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&evPacked_, hipEventDisableTiming));
    for (int i = 0; i < 4; ++i) HIP_CHECK(hipEventCreate(&evPhase_[i]));
    HIP_CHECK(hipMalloc(&wgt_, filter.size() * sizeof(float)));
    HIP_CHECK(hipMemcpy(wgt_, filter.data(), filter.size() * sizeof(float),
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&diffDev_, sizeof(float)));
    // Fused double-iteration policy: halves HBM traffic per iteration —
    // on by default for the bandwidth-bound fp16/fp32 storage paths
    // (the u8 path is issue-bound; measured A/B in profiles/).
    // PCONV_FUSE=0/1 overrides.
    const char* fe = getenv("PCONV_FUSE");
    if (fe) {
      fusePairs_ = atoi(fe) != 0;
    } else {
      fusePairs_ = dt_ != DType::U8;
    }
    // Exchange-overlap depth: how many iterations' INTERIORS launch on
    // the compute stream while the chunk's halo exchange is in flight
    // on the comm stream (their border bands replay after the exchange
    // event). Depth 1 is the classic interior/border split; deeper
    // overlap covers an exchange longer than one interior iteration
    // (~19 us at the 8-GPU rank shape) at the cost of 4 extra thin band
    // launches per depth step. Default 2; PCONV_OVERLAP_DEPTH overrides.
    const char* oe = getenv("PCONV_OVERLAP_DEPTH");
    overlapDepth_ = oe ? std::max(1, atoi(oe)) : 2;
  }

  ~ConvEngine() {
    for (auto& b : blocks_) {
      if (b.sendW) (void)hipFree(b.sendW);
      if (b.sendE) (void)hipFree(b.sendE);
      if (b.recvW) (void)hipFree(b.recvW);
      if (b.recvE) (void)hipFree(b.recvE);
      if (b.sendN) (void)hipFree(b.sendN);
      if (b.sendS) (void)hipFree(b.sendS);
    }
    if (wgt_) (void)hipFree(wgt_);
    if (diffDev_) (void)hipFree(diffDev_);
    if (comm_nccl_) (void)ncclCommDestroy(comm_nccl_);
    (void)hipEventDestroy(evComm_);
    (void)hipEventDestroy(evCompute_);
    (void)hipEventDestroy(evPacked_);
    for (int i = 0; i < 4; ++i) (void)hipEventDestroy(evPhase_[i]);
    (void)hipStreamDestroy(compute_);
    (void)hipStreamDestroy(comm_);
  }

  int add_block(uintptr_t src, uintptr_t dst, const BlockGeom& g) {
    DeviceBlock b;
    b.g = g;
    b.src = reinterpret_cast<void*>(src);
    b.dst = reinterpret_cast<void*>(dst);
    const size_t strip = (size_t)g.h * g.haloPx * g.S * itemsize(dt_);
    const size_t rowsB = (size_t)g.haloPx * g.pitchE * itemsize(dt_);
    if (g.nbr[NB_W] >= 0) {
      HIP_CHECK(hipMalloc(&b.sendW, strip));
      HIP_CHECK(hipMalloc(&b.recvW, strip));
    }
    if (g.nbr[NB_E] >= 0) {
      HIP_CHECK(hipMalloc(&b.sendE, strip));
      HIP_CHECK(hipMalloc(&b.recvE, strip));
    }
    if (g.nbr[NB_N] >= 0) HIP_CHECK(hipMalloc(&b.sendN, rowsB));
    if (g.nbr[NB_S] >= 0) HIP_CHECK(hipMalloc(&b.sendS, rowsB));
    blocks_.push_back(b);
    // fused integer-dot pairs (conv_u8dot2_k) halve HBM traffic but the
    // redundant iteration-A apron costs more than the bandwidth saved —
    // measured 1256 vs 1658 Gpx-it/s at 16384² (the dot path is
    // issue-bound, not BW-bound). Off by default; PCONV_DOT_FUSE=1
    // selects it for A/B.
    const char* de = getenv("PCONV_DOT_FUSE");
    dotFuse_ = de ? atoi(de) != 0 : false;
    return (int)blocks_.size() - 1;
  }

  void init_nccl(const std::string& uid, int rank, int world) {
    if (blocks_.size() != 1)
      throw std::runtime_error("NCCL mode expects exactly one local block");
    ncclUniqueId id;
    if (uid.size() < sizeof(id)) throw std::runtime_error("bad nccl uid size");
    std::memcpy(&id, uid.data(), sizeof(id));
    NCCL_CHECK(ncclCommInitRank(&comm_nccl_, world, id, rank));
  }

  static std::string get_unique_id() {
    ncclUniqueId id;
    NCCL_CHECK(ncclGetUniqueId(&id));
    return std::string(reinterpret_cast<char*>(&id), sizeof(id));
  }

  // Run `iters` iterations in chunks of <= chunkT (exchange depth r*T per
  // chunk). Fully asynchronous; blocks until the device finishes. With
  // phase timing enabled (per-phase tracing), hipEvent
  // pairs around each chunk's exchange and compute accumulate into
  // exchangeMs_/computeMs_ (adds one device sync per chunk — keep it off
  // in benchmark runs).
  void run(int iters, int chunkT) {
    const int r = k_ / 2;
    int left = iters;
    while (left > 0) {
      const int T = std::min(chunkT, left);
      const int d = r * T;
      if (phaseTiming_) HIP_CHECK(hipEventRecord(evPhase_[0], comm_));
      int t0 = 0;
      if (r > 0 && d > 0 && any_neighbor()) {
        // Overlap (BASELINE.json north-star: halo exchange "overlapped
        // with interior-tile compute on a second HIP stream"): the
        // INTERIORS of iterations 0..m-1 — regions shrunk so their input
        // windows never touch exchanged data — launch on the compute
        // stream concurrently with the exchange on the comm stream;
        // each iteration's four border BANDS replay afterwards against
        // that iteration's historical buffer pair (band j reads band
        // j-1's output and interior j's ring — both already ordered on
        // the compute stream, so the schedule is race-free by stream
        // order alone).
        const int m = std::min(overlapDepth_, T);
        HIP_CHECK(hipEventRecord(evCompute_, compute_));
        HIP_CHECK(hipStreamWaitEvent(comm_, evCompute_, 0));
        // exchange enqueues FIRST (host order) and operates on the
        // chunk-start buffers; it only STARTS after evCompute_, so on
        // the GPU it runs concurrently with the interiors below
        if (comm_nccl_) {
          exchange_nccl(d);
        } else {
          exchange_local(d);
        }
        hist_.clear();
        for (int j = 0; j < m; ++j) {
          if (j == 1) {
            // interiors j>=1 write buffers the exchange's capture phase
            // reads (the double-buffer flip aliases them onto the
            // chunk-start src): wait until every send has left src
            HIP_CHECK(hipStreamWaitEvent(compute_, evPacked_, 0));
          }
          for (auto& b : blocks_)
            launch_region(b, interior_region(b.g, (j + 1) * r));
          for (auto& b : blocks_) {
            hist_.push_back({b.src, b.dst});
            std::swap(b.src, b.dst);
          }
          swapped_ = !swapped_;
        }
        if (phaseTiming_) HIP_CHECK(hipEventRecord(evPhase_[1], comm_));
        HIP_CHECK(hipEventRecord(evComm_, comm_));
        HIP_CHECK(hipStreamWaitEvent(compute_, evComm_, 0));
        for (int j = 0; j < m; ++j) {
          const int ext = d - (j + 1) * r;
          for (size_t bi = 0; bi < blocks_.size(); ++bi) {
            auto& b = blocks_[bi];
            const auto& hp = hist_[(size_t)j * blocks_.size() + bi];
            Region full = region_for(b.g, ext);
            Region in = interior_region(b.g, (j + 1) * r);
            // N / S full-width strips, W / E strips between them
            Region strips[4] = {
                {full.py0, in.py0, full.ex0, full.ex1},
                {in.py1, full.py1, full.ex0, full.ex1},
                {in.py0, in.py1, full.ex0, in.ex0},
                {in.py0, in.py1, in.ex1, full.ex1},
            };
            for (const Region& s : strips)
              launch_one(b.g, hp.first, hp.second, s);
          }
        }
        t0 = m;
      }
      if (phaseTiming_ && t0 == 0)
        HIP_CHECK(hipEventRecord(evPhase_[1], comm_));  // no exchange
      if (phaseTiming_) HIP_CHECK(hipEventRecord(evPhase_[2], compute_));
      const bool canFuse =
          fusePairs_ && !useMfma_ && r >= 1 && r <= 2 && (S_ == 1 || S_ == 3);
      const bool canDotFuse = dotFuse_ && dotShift_ > 0 && !sepOn_;
      for (int t = t0; t < T;) {
        if ((canFuse || canDotFuse) && t + 1 < T) {
          // fused pair: iteration t into LDS mid, iteration t+1 to global
          bool ok = true;
          for (auto& b : blocks_) {
            const Region rm = region_for(b.g, d - (t + 1) * r);
            const Region ro = region_for(b.g, d - (t + 2) * r);
            if (canDotFuse) {
              launch_conv_u8dot2(b.src, b.dst, dotW_, dotShift_, b.g, rm, ro,
                                 compute_);
            } else {
              ok &= launch_conv2(dt_, S_, k_, b.src, b.dst,
                                 (const float*)wgt_, b.g, rm, ro, compute_);
            }
          }
          if (ok) {
            for (auto& b : blocks_) std::swap(b.src, b.dst);
            swapped_ = !swapped_;
            t += 2;
            continue;
          }
        }
        const int ext = d - (t + 1) * r;
        for (auto& b : blocks_) launch_region(b, region_for(b.g, ext));
        for (auto& b : blocks_) std::swap(b.src, b.dst);
        swapped_ = !swapped_;
        ++t;
      }
      left -= T;
      if (phaseTiming_) {
        HIP_CHECK(hipEventRecord(evPhase_[3], compute_));
        HIP_CHECK(hipEventSynchronize(evPhase_[3]));
        float ms = 0.f;
        HIP_CHECK(hipEventElapsedTime(&ms, evPhase_[0], evPhase_[1]));
        exchangeMs_ += ms;
        HIP_CHECK(hipEventElapsedTime(&ms, evPhase_[2], evPhase_[3]));
        computeMs_ += ms;
        chunks_ += 1;
      }
    }
    HIP_CHECK(hipStreamSynchronize(compute_));
    HIP_CHECK(hipStreamSynchronize(comm_));
    HIP_CHECK(hipGetLastError());
  }

  void set_phase_timing(bool on) { phaseTiming_ = on; }

  // which compute path run() will take (observability: lands in the
  // bench JSON so a measurement names the kernel that produced it)
  std::string kernel_path() const {
    if (sepOn_) return "u8sep";
    if (dotShift_ > 0) return dotFuse_ ? "u8dot-fused" : "u8dot";
    if (useMfma_) return "mfma-f16";
    if (fusePairs_ && k_ / 2 >= 1 && k_ / 2 <= 2 && (S_ == 1 || S_ == 3))
      return "valu-fused2";
    return "valu";
  }
  // (exchange_ms, compute_ms, chunks) accumulated since construction
  std::vector<double> phase_times() const {
    return {exchangeMs_, computeMs_, (double)chunks_};
  }

  // max |src - dst| over owned regions of all local blocks (convergence
  // test). Synchronous.
  float diff_max() {
    HIP_CHECK(hipMemsetAsync(diffDev_, 0, sizeof(float), compute_));
    for (auto& b : blocks_)
      launch_diff_max(dt_, b.src, b.dst, b.g, (float*)diffDev_, compute_);
    float out = 0.f;
    HIP_CHECK(hipMemcpyAsync(&out, diffDev_, sizeof(float),
                             hipMemcpyDeviceToHost, compute_));
    HIP_CHECK(hipStreamSynchronize(compute_));
    return out;
  }

  // true if the latest result lives in the tensor originally passed
  // as dst (odd total iteration count)
  bool swapped() const { return swapped_; }

  void synchronize() {
    HIP_CHECK(hipStreamSynchronize(compute_));
    HIP_CHECK(hipStreamSynchronize(comm_));
  }

 private:
  bool any_neighbor() const {
    for (const auto& b : blocks_)
      for (int i = 0; i < 4; ++i)
        if (b.g.nbr[i] >= 0) return true;
    return false;
  }

  // iteration-j interior: the owned block shrunk by `depth` on every
  // side — its input windows never touch data the in-flight exchange
  // will deliver (empty if the block is too thin — the border bands
  // then cover everything)
  Region interior_region(const BlockGeom& g, int depth) const {
    Region rg;
    rg.py0 = g.haloPx + depth;
    rg.py1 = std::max(g.haloPx + g.h - depth, rg.py0);
    rg.ex0 = (g.haloPx + depth) * g.S;
    rg.ex1 = std::max((g.haloPx + g.wPx - depth) * g.S, rg.ex0);
    return rg;
  }

  // kernel-path selection on explicit buffers: integer separable / dot
  // fast paths when eligible, else MFMA/VALU via launch_conv
  void launch_path(const BlockGeom& g, void* src, void* dst,
                   const Region& rg) {
    if (sepOn_) {
      launch_conv_u8sep(src, dst, sepW_, k_, g, rg, compute_);
    } else if (dotShift_ > 0) {
      launch_conv_u8dot(src, dst, dotW_, dotShift_, g, rg, compute_);
    } else {
      launch_conv(dt_, S_, k_, src, dst, (const float*)wgt_, g, rg, useMfma_,
                  compute_);
    }
  }

  void launch_region(DeviceBlock& b, const Region& rg) {
    launch_path(b.g, b.src, b.dst, rg);
  }

  // border-band launch: thin strips route to the element-indexed
  // generic kernel — the 1024-element tiled kernels waste ~30x on
  // masked lanes there (bit-compatible: fp32 fma in the same tap order
  // with the same store rounding; exact classes stay exact)
  void launch_one(const BlockGeom& g, void* src, void* dst,
                  const Region& rg) {
    if (rg.py1 <= rg.py0 || rg.ex1 <= rg.ex0) return;
    static const bool noStrip = [] {
      const char* e = getenv("PCONV_STRIP_ROUTE");
      return e && atoi(e) == 0;
    }();
    const bool thin =
        !noStrip && ((rg.ex1 - rg.ex0) < 256 || (rg.py1 - rg.py0) < 16);
    if (thin) {
      launch_conv_strip(dt_, S_, k_, src, dst, (const float*)wgt_, g, rg,
                        compute_);
    } else {
      launch_path(g, src, dst, rg);
    }
  }

  Region region_for(const BlockGeom& g, int ext) const {
    Region r;
    r.py0 = g.haloPx - (g.nbr[NB_N] >= 0 ? ext : 0);
    r.py1 = g.haloPx + g.h + (g.nbr[NB_S] >= 0 ? ext : 0);
    const int pxLo = g.haloPx - (g.nbr[NB_W] >= 0 ? ext : 0);
    const int pxHi = g.haloPx + g.wPx + (g.nbr[NB_E] >= 0 ? ext : 0);
    r.ex0 = pxLo * g.S;
    r.ex1 = pxHi * g.S;
    return r;
  }

  // --- exchange ------------------------------------------------------------

  char* elem(DeviceBlock& b, void* base, int row, int colPx) const {
    return (char*)base +
           ((size_t)row * b.g.pitchE + (size_t)colPx * b.g.S) * itemsize(dt_);
  }

  void copy2d(void* dst, size_t dpitch, const void* src, size_t spitch,
              size_t widthB, size_t rows) {
    // pack kernel instead of hipMemcpy2DAsync: the SDMA strided path
    // costs ~41 us for a 32 B x 16384-row strip vs ~5 us kernel
    // (bench/exchange_cost.py); this sits on the exchange critical path
    // every chunk. PCONV_SDMA_PACK=1 restores the SDMA path for A/B.
    static const bool sdma = [] {
      const char* e = getenv("PCONV_SDMA_PACK");
      return e && atoi(e) != 0;
    }();
    if (sdma) {
      HIP_CHECK(hipMemcpy2DAsync(dst, dpitch, src, spitch, widthB, rows,
                                 hipMemcpyDeviceToDevice, comm_));
    } else {
      launch_copy2d(dst, dpitch, src, spitch, widthB, rows, comm_);
      HIP_CHECK(hipGetLastError());
    }
  }

  // LOCAL mode: neighbors are local block indices; D2D copies.
  void exchange_local(int d) {
    const int is = itemsize(dt_);
    // Phase A: E/W column strips (owned rows only)
    for (auto& b : blocks_) {
      if (b.g.nbr[NB_E] < 0) continue;
      DeviceBlock& e = blocks_[b.g.nbr[NB_E]];
      const int H = b.g.haloPx;
      const size_t wB = (size_t)d * b.g.S * is;
      const size_t bp = (size_t)b.g.pitchE * is;
      const size_t ep = (size_t)e.g.pitchE * is;
      // my east strip -> east's west halo
      copy2d(elem(e, e.src, H, H - d), ep, elem(b, b.src, H, H + b.g.wPx - d),
             bp, wB, b.g.h);
      // east's west strip -> my east halo
      copy2d(elem(b, b.src, H, H + b.g.wPx), bp, elem(e, e.src, H, H), ep, wB,
             b.g.h);
    }
    // Phase B: N/S full-pitch row strips (contiguous)
    for (auto& b : blocks_) {
      if (b.g.nbr[NB_S] < 0) continue;
      DeviceBlock& s = blocks_[b.g.nbr[NB_S]];
      const int H = b.g.haloPx;
      const size_t rowB = (size_t)b.g.pitchE * is;
      // my bottom d rows -> south's north halo
      HIP_CHECK(hipMemcpyAsync(elem(s, s.src, H - d, 0),
                               elem(b, b.src, H + b.g.h - d, 0), rowB * d,
                               hipMemcpyDeviceToDevice, comm_));
      // south's top d rows -> my south halo
      HIP_CHECK(hipMemcpyAsync(elem(b, b.src, H + b.g.h, 0),
                               elem(s, s.src, H, 0), rowB * d,
                               hipMemcpyDeviceToDevice, comm_));
    }
    // all reads of src complete at this point in comm-stream order: the
    // overlap pyramid's interiors j>=1 (which overwrite src edge
    // strips) gate on this event
    HIP_CHECK(hipEventRecord(evPacked_, comm_));
  }

  // RCCL mode: one block, neighbors are global ranks.
  //
  // All send data leaves `src` in the capture phase below, BEFORE any
  // network round, so evPacked_ lets the overlap pyramid's interiors
  // j>=1 overwrite src edge strips while the exchange is in flight
  // (without staging, iteration 1's interior raced the sends —
  // caught by the r02 fuzz campaign). The N/S rows' E/W-halo corner
  // columns only become valid after phase A; they are patched into the
  // staged rows FROM THE RECV STAGING (never from src) — identical
  // bytes to what phase A unpacks into src.
  void exchange_nccl(int d) {
    DeviceBlock& b = blocks_[0];
    const int is = itemsize(dt_);
    const int H = b.g.haloPx;
    const size_t stripB = (size_t)b.g.h * d * b.g.S * is;  // phase-A message
    const size_t wB = (size_t)d * b.g.S * is;              // strip row bytes
    const size_t bp = (size_t)b.g.pitchE * is;
    const size_t rowsB = (size_t)d * bp;
    const int W = b.g.nbr[NB_W], E = b.g.nbr[NB_E];
    const int N = b.g.nbr[NB_N], Sn = b.g.nbr[NB_S];

    // capture phase: E/W column strips and N/S row strips -> staging
    if (W >= 0)
      copy2d(b.sendW, wB, elem(b, b.src, H, H), bp, wB, b.g.h);
    if (E >= 0)
      copy2d(b.sendE, wB, elem(b, b.src, H, H + b.g.wPx - d), bp, wB, b.g.h);
    if (N >= 0)
      HIP_CHECK(hipMemcpyAsync(b.sendN, elem(b, b.src, H, 0), rowsB,
                               hipMemcpyDeviceToDevice, comm_));
    if (Sn >= 0)
      HIP_CHECK(hipMemcpyAsync(b.sendS, elem(b, b.src, H + b.g.h - d, 0),
                               rowsB, hipMemcpyDeviceToDevice, comm_));
    HIP_CHECK(hipEventRecord(evPacked_, comm_));

    // Phase A: E/W grouped send/recv, unpack into src halo columns.
    if (W >= 0 || E >= 0) {
      NCCL_CHECK(ncclGroupStart());
      if (W >= 0) {
        NCCL_CHECK(ncclSend(b.sendW, stripB, ncclChar, W, comm_nccl_, comm_));
        NCCL_CHECK(ncclRecv(b.recvW, stripB, ncclChar, W, comm_nccl_, comm_));
      }
      if (E >= 0) {
        NCCL_CHECK(ncclSend(b.sendE, stripB, ncclChar, E, comm_nccl_, comm_));
        NCCL_CHECK(ncclRecv(b.recvE, stripB, ncclChar, E, comm_nccl_, comm_));
      }
      NCCL_CHECK(ncclGroupEnd());
      if (W >= 0)
        copy2d(elem(b, b.src, H, H - d), bp, b.recvW, wB, wB, b.g.h);
      if (E >= 0)
        copy2d(elem(b, b.src, H, H + b.g.wPx), bp, b.recvE, wB, wB, b.g.h);
    }

    // corner patch: the W/E d-wide column ranges of the staged N/S rows
    // get the just-received phase-A data (corner propagation of the
    // two-phase scheme), sourced from recvW/recvE staging.
    auto patch = [&](void* sendRows, size_t recvRowOff) {
      if (W >= 0)
        copy2d((char*)sendRows + (size_t)(H - d) * b.g.S * is, bp,
               (const char*)b.recvW + recvRowOff * wB, wB, wB, d);
      if (E >= 0)
        copy2d((char*)sendRows + (size_t)(H + b.g.wPx) * b.g.S * is, bp,
               (const char*)b.recvE + recvRowOff * wB, wB, wB, d);
    };
    if (N >= 0) patch(b.sendN, 0);
    if (Sn >= 0) patch(b.sendS, (size_t)b.g.h - d);

    // Phase B: N/S full-pitch rows — sends from staging, recvs direct.
    if (N >= 0 || Sn >= 0) {
      NCCL_CHECK(ncclGroupStart());
      if (N >= 0) {
        NCCL_CHECK(ncclSend(b.sendN, rowsB, ncclChar, N, comm_nccl_, comm_));
        NCCL_CHECK(ncclRecv(elem(b, b.src, H - d, 0), rowsB, ncclChar, N,
                            comm_nccl_, comm_));
      }
      if (Sn >= 0) {
        NCCL_CHECK(ncclSend(b.sendS, rowsB, ncclChar, Sn, comm_nccl_, comm_));
        NCCL_CHECK(ncclRecv(elem(b, b.src, H + b.g.h, 0), rowsB, ncclChar, Sn,
                            comm_nccl_, comm_));
      }
      NCCL_CHECK(ncclGroupEnd());
    }
  }

  DType dt_;
  int k_;
  int S_;
  std::vector<DeviceBlock> blocks_;
  hipStream_t compute_ = nullptr;
  hipStream_t comm_ = nullptr;
  hipEvent_t evComm_ = nullptr;
  hipEvent_t evCompute_ = nullptr;
  hipEvent_t evPacked_ = nullptr;
  void* wgt_ = nullptr;
  void* diffDev_ = nullptr;
  ncclComm_t comm_nccl_ = nullptr;
  bool swapped_ = false;
  bool useMfma_ = false;
  bool fusePairs_ = false;
  bool sepOn_ = false;
  int sepW_[11] = {};
  int dotShift_ = 0;
  bool dotFuse_ = false;
  unsigned dotW_[9] = {};
  int overlapDepth_ = 2;
  std::vector<std::pair<void*, void*>> hist_;  // per-overlap-iter buffers
  bool phaseTiming_ = false;
  hipEvent_t evPhase_[4] = {};
  double exchangeMs_ = 0.0;
  double computeMs_ = 0.0;
  long chunks_ = 0;
};

}  // namespace pconv
