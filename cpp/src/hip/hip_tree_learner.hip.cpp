/*!
 * migbm HIP tree learner — full-GPU leaf-wise histogram learner for MI355X (gfx950).
 *
 * Fresh CDNA4-first design (capability target: the reference's CUDA single-GPU learner,
 * src/treelearner/cuda/* — see SURVEY.md §2.3; NOT a port). Key design points:
 *  - binned matrix resident in HBM twice: row-major uint8 (16B-aligned rows, histogram
 *    build with uint4 vector loads) and column-major (partition decisions, coalesced)
 *  - ConstructHistogram: LDS histograms privatized per lane-group (up to 4 copies keyed
 *    by lane%4) to break the 64-wide same-bin DS-atomic serialization that dominates on
 *    homogeneous leaves; feature-partitioned to the LDS budget; grid-stride rows
 *  - ONE host<->device sync per split: all leaf bookkeeping (begin/count/slot/stats)
 *    lives in device arrays maintained by a tiny finalize kernel; the histogram kernel
 *    self-selects the smaller child from global counts; the only readback is the
 *    winning split record (pinned staging)
 *  - best-split: one wave64 per (child leaf, feature), shfl inclusive prefix scan over
 *    bins, both missing directions, fp64 gain math identical to the CPU oracle
 *  - partition: single-pass wave-aggregated atomic-counter partition (left packed
 *    ascending, right packed descending) + copy-back
 *  - multi-GPU: one process per GPU; RCCL allreduce of the smaller-child histogram,
 *    root sums and child counts over xGMI, stream-ordered (adds no host sync)
 */
#include <hip/hip_runtime.h>
#include <hip/hip_cooperative_groups.h>
#include <rccl/rccl.h>

#include "migbm/tree_learner.h"
#include "migbm/objective.h"
#include "migbm/network.h"

#include <algorithm>
#include <condition_variable>
#include <mutex>
#include <thread>

namespace migbm {

#define HIP_OK(call)                                                                  \
  do {                                                                                \
    hipError_t _e = (call);                                                           \
    if (_e != hipSuccess)                                                             \
      ::migbm::Log::Fatal("HIP error %s at %s:%d", hipGetErrorString(_e), __FILE__,   \
                          __LINE__);                                                  \
  } while (0)

#define NCCL_OK(call)                                                                  \
  do {                                                                                 \
    ncclResult_t _e = (call);                                                          \
    if (_e != ncclSuccess)                                                             \
      ::migbm::Log::Fatal("RCCL error %s at %s:%d", ncclGetErrorString(_e), __FILE__,  \
                          __LINE__);                                                   \
  } while (0)

// ------------------------------------------------------------------ GPU comm singleton
/*! Transport-agnostic inter-GPU collectives. Two transports:
 *  - RCCL (ncclCommInitRank via LGBM_GPUNetworkInit): the production path, one
 *    process per GPU over xGMI, stream-ordered (no host syncs).
 *  - host seam (migbm::Network, injected allgather): correctness transport used
 *    when several ranks share one GPU (RCCL refuses duplicate devices) — this is
 *    how the multi-rank device code path is proven on a single-GPU box. Each call
 *    syncs the stream and bounces through host memory.
 *  Both run the IDENTICAL device-side code; only the reduction transport differs. */
/*! in-process clique: barrier-framed host-bounce collectives between the
 *  single-process num_gpu worker threads (one per device context). Used when
 *  RCCL cannot (several shards share one physical GPU on a test box) and as the
 *  universal fallback; distinct devices use a real ncclCommInitAll clique. */
struct InProcClique {
  explicit InProcClique(int w) : world(w) {}
  const int world;

  void Barrier() {
    std::unique_lock<std::mutex> lk(mu_);
    const uint64_t ph = phase_;
    if (++arrived_ == world) {
      arrived_ = 0;
      ++phase_;
      cv_.notify_all();
    } else {
      cv_.wait(lk, [&] { return phase_ != ph; });
    }
  }

  template <typename T>
  void AllReduceHost(T* h, size_t n, int rank) {
    Barrier();
    if (rank == 0 && buf_.size() < static_cast<size_t>(world) * n * sizeof(T))
      buf_.resize(static_cast<size_t>(world) * n * sizeof(T));
    Barrier();
    T* slots = reinterpret_cast<T*>(buf_.data());
    memcpy(slots + static_cast<size_t>(rank) * n, h, n * sizeof(T));
    Barrier();
    for (size_t i = 0; i < n; ++i) {
      T acc = 0;
      for (int r = 0; r < world; ++r) acc += slots[static_cast<size_t>(r) * n + i];
      h[i] = acc;
    }
    Barrier();
  }

  void AllGatherHost(const void* in, void* out, size_t bytes, int rank) {
    Barrier();
    if (rank == 0 && buf_.size() < static_cast<size_t>(world) * bytes)
      buf_.resize(static_cast<size_t>(world) * bytes);
    Barrier();
    memcpy(buf_.data() + static_cast<size_t>(rank) * bytes, in, bytes);
    Barrier();
    memcpy(out, buf_.data(), static_cast<size_t>(world) * bytes);
    Barrier();
  }

 private:
  std::mutex mu_;
  std::condition_variable cv_;
  int arrived_ = 0;
  uint64_t phase_ = 0;
  std::vector<char> buf_;
};

struct GpuComm {
  ncclComm_t comm = nullptr;
  int world = 1;
  int rank = 0;
  InProcClique* clique = nullptr;  // single-process num_gpu transport
  bool active() const {
    return comm != nullptr || clique != nullptr || Network::is_distributed();
  }
  bool rccl() const { return comm != nullptr; }
  int World() const {
    if (rccl()) return world;
    if (clique) return clique->world;
    return Network::num_machines();
  }
  int Rank() const {
    if (rccl()) return rank;
    if (clique) return rank;
    return Network::rank();
  }

  void AllReduce(float* d, size_t n, hipStream_t s) {
    if (rccl()) {
      NCCL_OK(ncclAllReduce(d, d, n, ncclFloat32, ncclSum, comm, s));
    } else {
      HostBounce<float>(d, n, s);
    }
  }
  void AllReduce(double* d, size_t n, hipStream_t s) {
    if (rccl()) {
      NCCL_OK(ncclAllReduce(d, d, n, ncclFloat64, ncclSum, comm, s));
    } else {
      HostBounce<double>(d, n, s);
    }
  }
  void AllReduce(int64_t* d, size_t n, hipStream_t s) {
    if (rccl()) {
      NCCL_OK(ncclAllReduce(d, d, n, ncclInt64, ncclSum, comm, s));
    } else {
      HostBounce<int64_t>(d, n, s);
    }
  }
  /*! per-rank-owned block reduce: rank r ends with the globally-summed
   *  [off[r], off[r]+cnt[r]) slice; other slices are left partial (garbage).
   *  RCCL: grouped ncclReduce, one per root. Host seam: plain allreduce (a
   *  correct superset — owned slices are what the kernels read). */
  template <typename T>
  void ReduceBlocks(T* d, const std::vector<size_t>& off, const std::vector<size_t>& cnt,
                    hipStream_t s) {
    if (rccl()) {
      const ncclDataType_t dt = sizeof(T) == 8 ? ncclFloat64 : ncclFloat32;
      NCCL_OK(ncclGroupStart());
      for (int r = 0; r < world; ++r) {
        if (cnt[r] == 0) continue;
        NCCL_OK(ncclReduce(d + off[r], d + off[r], cnt[r], dt, ncclSum, r, comm, s));
      }
      NCCL_OK(ncclGroupEnd());
    } else {
      HostBounce<T>(d, off.back() + cnt.back(), s);
    }
  }
  /*! byte allgather: world * bytes_per_rank into d_out (rank-major). */
  void AllGather(const void* d_in, void* d_out, size_t bytes_per_rank, hipStream_t s);

  static GpuComm& Get() {
    static GpuComm c;
    return c;
  }

 private:
  template <typename T>
  void HostBounce(T* d, size_t n, hipStream_t s) {
    if (hipStreamSynchronize(s) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: stream sync failed");
    staging_.resize(n * sizeof(T));
    T* h = reinterpret_cast<T*>(staging_.data());
    if (hipMemcpy(h, d, n * sizeof(T), hipMemcpyDeviceToHost) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: D2H failed");
    if (clique) clique->AllReduceHost(h, n, rank);
    else Network::AllreduceSum(h, n);
    if (hipMemcpy(d, h, n * sizeof(T), hipMemcpyHostToDevice) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: H2D failed");
  }
  std::vector<char> staging_;
};

void GpuComm::AllGather(const void* d_in, void* d_out, size_t bytes_per_rank, hipStream_t s) {
  if (rccl()) {
    NCCL_OK(ncclAllGather(d_in, d_out, bytes_per_rank, ncclInt8, comm, s));
  } else {
    if (hipStreamSynchronize(s) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: stream sync failed");
    const int w = World();
    std::vector<char> h_in(bytes_per_rank), h_out(bytes_per_rank * w);
    if (hipMemcpy(h_in.data(), d_in, bytes_per_rank, hipMemcpyDeviceToHost) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: D2H failed");
    if (clique)
      clique->AllGatherHost(h_in.data(), h_out.data(), bytes_per_rank, rank);
    else
      Network::Allgather(h_in.data(), static_cast<int>(bytes_per_rank), h_out.data());
    if (hipMemcpy(d_out, h_out.data(), bytes_per_rank * w, hipMemcpyHostToDevice) != hipSuccess)
      Log::Fatal("GpuComm host-bounce: H2D failed");
  }
}

namespace hipk {

struct SplitRec {
  double gain;
  double left_g, left_h;
  double left_out, right_out;
  // bin-level subset for categorical sorted-subset splits, 256 bins max
  // (4x64-bit words); all-zero = one-hot/numeric
  unsigned long long cat_mask[4];
  int left_cnt, right_cnt;  // hessian-derived approx (host launch-sizing hints)
  int feature;
  int bin;
  int default_left;
  int valid;
};

struct LogEntry {
  SplitRec rec;
  int leaf;
  int pad;
};

struct LeafStat {
  double sum_g, sum_h;
  double parent_out;  // hessian-weighted parent output (path smoothing)
  int cnt;    // exact GLOBAL row count (allreduced in multi-GPU)
  int depth;  // leaf depth (max_depth gating + monotone_penalty decay)
};

struct GainParams {
  double l1, l2, mds;
  double min_hess, min_gain_to_split;
  int min_data;
  // sync-free per-split sampling (device hash RNG keyed on split counter):
  float bynode_frac;   // feature_fraction_bynode (1.0 = off)
  int extra_trees;     // 1 = evaluate one hashed random threshold per feature
  uint32_t rng_seed;   // per-tree seed component
  double smooth;       // path_smooth (0 = off; smoothing uses LeafStat.parent_out)
  // categorical scan (parity: FindBestThresholdCategorical CPU oracle)
  double cat_l2, cat_smooth;
  int max_cat_to_onehot, max_cat_threshold;
  int n_interaction_groups;  // 0 = unconstrained
  // distributed feature ownership: this rank scans only features [own_fb, own_fe)
  // (reduce-scatter mode divides both wire bytes and the gain scan by world)
  int own_fb, own_fe;
  int max_depth;         // 0/neg = unbounded
  double mono_penalty;   // monotone_penalty depth-decay factor (0 = off)
  double cegb_tradeoff;  // cost-effective gradient boosting (0 = off)
  double cegb_split_pen;
};

__device__ __forceinline__ uint32_t d_hash3(uint32_t a, uint32_t b, uint32_t c) {
  uint32_t x = a * 0x9E3779B1u ^ b * 0x85EBCA77u ^ c * 0xC2B2AE3Du;
  x ^= x >> 16;
  x *= 0x7FEB352Du;
  x ^= x >> 15;
  x *= 0x846CA68Bu;
  x ^= x >> 16;
  return x;
}

struct FeatMeta {
  int bin_off;
  int num_bin;
  int num_numeric_bin;
  int nan_bin;
  int is_cat;
};

__device__ __forceinline__ double d_thl1(double s, double l1) {
  double r = fabs(s) - l1;
  r = r > 0.0 ? r : 0.0;
  return s >= 0.0 ? r : -r;
}
/*! cost-effective gradient boosting gain penalty (CPU-oracle parity:
 *  SerialTreeLearner CEGB block). coupled[f] is zeroed once f is used. */
__device__ __forceinline__ void d_apply_cegb(SplitRec& rec, int f, int num_data,
                                             const GainParams& p,
                                             const float* __restrict__ coupled,
                                             const float* __restrict__ lazy) {
  if (p.cegb_tradeoff <= 0.0 || !rec.valid) return;
  double pen = p.cegb_split_pen;
  if (coupled != nullptr) pen += coupled[f];
  if (lazy != nullptr) pen += static_cast<double>(lazy[f]) * num_data;
  rec.gain -= p.cegb_tradeoff * pen;
  if (rec.gain <= 0.0) rec.valid = 0;
}

__device__ __forceinline__ double d_leaf_out(double g, double h, const GainParams& p) {
  double r = -d_thl1(g, p.l1) / (h + p.l2);
  if (p.mds > 0.0 && fabs(r) > p.mds) r = r > 0 ? p.mds : -p.mds;
  return r;
}
__device__ __forceinline__ double d_gain_out(double g, double h, double out,
                                             const GainParams& p) {
  double s = d_thl1(g, p.l1);
  return -(2.0 * s * out + (h + p.l2) * out * out);
}
__device__ __forceinline__ double d_leaf_out_l2(double g, double h, const GainParams& p,
                                                double l2v) {
  double r = -d_thl1(g, p.l1) / (h + l2v);
  if (p.mds > 0.0 && fabs(r) > p.mds) r = r > 0 ? p.mds : -p.mds;
  return r;
}
__device__ __forceinline__ double d_gain_out_l2(double g, double h, double out,
                                                const GainParams& p, double l2v) {
  double s = d_thl1(g, p.l1);
  return -(2.0 * s * out + (h + l2v) * out * out);
}
__device__ __forceinline__ double d_split_gain_l2(double gl, double hl, double gr, double hr,
                                                  const GainParams& p, double l2v) {
  if (p.mds <= 0.0) {
    const double sl = d_thl1(gl, p.l1), sr = d_thl1(gr, p.l1);
    return sl * sl / (hl + l2v) + sr * sr / (hr + l2v);
  }
  const double lo = d_leaf_out_l2(gl, hl, p, l2v), ro = d_leaf_out_l2(gr, hr, p, l2v);
  return d_gain_out_l2(gl, hl, lo, p, l2v) + d_gain_out_l2(gr, hr, ro, p, l2v);
}
/*! path smoothing (CPU-oracle parity): out*(n/a)/(n/a+1) + parent/(n/a+1) */
__device__ __forceinline__ double d_smooth_out(double out, double n, double parent_out,
                                               const GainParams& p) {
  if (p.smooth <= 0.0) return out;
  const double na = n / p.smooth;
  return out * na / (na + 1.0) + parent_out / (na + 1.0);
}
__device__ __forceinline__ double d_leaf_gain(double g, double h, const GainParams& p) {
  if (p.mds <= 0.0) {
    double s = d_thl1(g, p.l1);
    return s * s / (h + p.l2);
  }
  double out = d_leaf_out(g, h, p);
  return d_gain_out(g, h, out, p);
}
__device__ __forceinline__ double d_leaf_gain_sm(double g, double h, double n,
                                                 double parent_out, const GainParams& p) {
  if (p.smooth <= 0.0) return d_leaf_gain(g, h, p);
  const double out = d_smooth_out(d_leaf_out(g, h, p), n, parent_out, p);
  return d_gain_out(g, h, out, p);
}

/*! absmax of grad/hess over the used rows -> scales (double[2]). */
__global__ void k_grad_absmax(const uint32_t* __restrict__ idx, int cnt,
                              const float* __restrict__ g, const float* __restrict__ h,
                              float* __restrict__ out_max /* [2], pre-zeroed */) {
  float mg = 0, mh = 0;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < cnt; i += blockDim.x * gridDim.x) {
    const uint32_t r = idx[i];
    mg = fmaxf(mg, fabsf(g[r]));
    mh = fmaxf(mh, fabsf(h[r]));
  }
  for (int d = 32; d > 0; d >>= 1) {
    mg = fmaxf(mg, __shfl_down(mg, d));
    mh = fmaxf(mh, __shfl_down(mh, d));
  }
  if ((threadIdx.x & 63) == 0) {
    atomicMax(reinterpret_cast<int*>(&out_max[0]), __float_as_int(mg));
    atomicMax(reinterpret_cast<int*>(&out_max[1]), __float_as_int(mh));
  }
}

/*! stochastic-rounded quantization of (g,h) into a packed int32 (g<<16 | h).
 *  Range: g in [-levels, levels], h in [0, 2*levels]; scales derived from absmax. */
__global__ void k_grad_quantize(const float* __restrict__ g, const float* __restrict__ h,
                                int n, const float* __restrict__ absmax, int levels,
                                int use_stochastic, uint32_t seed,
                                int32_t* __restrict__ packed,
                                float* __restrict__ scales /* [2] out */) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i == 0) {
    scales[0] = absmax[0] > 0 ? absmax[0] / levels : 1.0f;
    scales[1] = absmax[1] > 0 ? absmax[1] / (2.0f * levels) : 1.0f;
  }
  if (i >= n) return;
  const float gs = absmax[0] > 0 ? absmax[0] / levels : 1.0f;
  const float hs = absmax[1] > 0 ? absmax[1] / (2.0f * levels) : 1.0f;
  float rg = 0.5f, rh = 0.5f;
  if (use_stochastic) {
    uint32_t x = (static_cast<uint32_t>(i) * 2654435761u) ^ seed;
    x ^= x >> 16;
    x *= 2246822519u;
    x ^= x >> 13;
    rg = (x & 0xFFFF) * (1.0f / 65536.0f);
    rh = ((x >> 16) & 0xFFFF) * (1.0f / 65536.0f);
  }
  int gq = static_cast<int>(floorf(g[i] / gs + rg));
  int hq = static_cast<int>(floorf(h[i] / hs + rh));
  gq = max(-levels, min(levels, gq));
  hq = max(0, min(2 * levels, hq));
  packed[i] = (gq << 16) | (hq & 0xFFFF);
}

/*! packed-int histogram: ONE ds_add per (row, feature); per-block partial sums are
 *  dequantized into the same fp32 global histogram at flush, so the best-split scan
 *  is unchanged. hi16 = sum of gq (signed), lo16 = sum of hq (non-negative; bounded
 *  by rows_per_block * 2*levels < 2^15 via grid sizing). */
template <int NCOPIES>
__global__ void k_hist_q(const uint8_t* __restrict__ rows, int stride,
                         const uint32_t* __restrict__ idx_base,
                         const int* __restrict__ leaf_begin, const int* __restrict__ leaf_cnt,
                         const LeafStat* __restrict__ stats, const int* __restrict__ leaf_slot,
                         const int* __restrict__ leafA_ptr, const int* __restrict__ counters,
                         int leafB_from_counters, const int32_t* __restrict__ packed,
                         const float* __restrict__ scales, const FeatMeta* __restrict__ fm,
                         int feat_begin, int feat_end, int part_bin_base, int part_bins,
                         float* __restrict__ hist_base, size_t slot_stride) {
  const int leafA = *leafA_ptr;
  if (leafA < 0) return;
  const int leafB = leafB_from_counters ? counters[0] - 1 : -1;
  int leaf = leafA;
  if (leafB >= 0 && stats[leafB].cnt < stats[leafA].cnt) leaf = leafB;
  const int begin = leaf_begin[leaf];
  const int cnt = leaf_cnt[leaf];
  const uint32_t* idx = idx_base + begin;
  float* ghist = hist_base + static_cast<size_t>(leaf_slot[leaf]) * slot_stride +
                 static_cast<size_t>(part_bin_base) * 2;

  constexpr int kBinStride = NCOPIES + 1;  // ints per bin (non-pow2 bank rotation)
  extern __shared__ int lhq[];
  __shared__ int loff[256];
  const int nfeat = feat_end - feat_begin;
  for (int i = threadIdx.x; i < nfeat; i += blockDim.x)
    loff[i] = fm[feat_begin + i].bin_off - part_bin_base;
  const int nelem = part_bins * kBinStride;
  for (int i = threadIdx.x; i < nelem; i += blockDim.x) lhq[i] = 0;
  __syncthreads();

  const int my_copy = threadIdx.x % NCOPIES;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  const int nthreads = blockDim.x * gridDim.x;
  const int c0 = feat_begin & ~15;
  for (int i = tid; i < cnt; i += nthreads) {
    const uint32_t r = idx[i];
    const int32_t gh = packed[r];
    const uint8_t* rp = rows + static_cast<size_t>(r) * stride;
    for (int c = c0; c < feat_end; c += 16) {
      const uint4 v = *reinterpret_cast<const uint4*>(rp + c);
      const uint32_t w[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int f = c + j;
        if (f < feat_begin || f >= feat_end) continue;
        const int b = (w[j >> 2] >> ((j & 3) * 8)) & 0xFF;
        atomicAdd(&lhq[(loff[f - feat_begin] + b) * kBinStride + my_copy], gh);
      }
    }
  }
  __syncthreads();
  const float gs = scales[0], hs = scales[1];
  for (int i = threadIdx.x; i < part_bins; i += blockDim.x) {
    int G = 0, H = 0;
#pragma unroll
    for (int cpy = 0; cpy < NCOPIES; ++cpy) {
      const int v = lhq[i * kBinStride + cpy];
      // v = G_c * 65536 + H_c with 0 <= H_c < 2^16 (bounds guaranteed by grid sizing)
      const int gq = v >> 16;
      G += gq;
      H += v - (gq << 16);
    }
    if (G != 0) atomicAdd(&ghist[i * 2], static_cast<float>(G) * gs);
    if (H != 0) atomicAdd(&ghist[i * 2 + 1], static_cast<float>(H) * hs);
  }
}

__global__ void k_iota(uint32_t* p, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = static_cast<uint32_t>(i);
}

// ------------------------------------------------------------------ histogram
/*! Histogram of the SMALLER of two candidate child leaves (leafB < 0 -> leafA).
 *  NCOPIES lane-group LDS copies (lane % NCOPIES) dilute same-bin DS-atomic conflicts:
 *  a fully homogeneous leaf serializes 64-wide without them (measured 92us for a
 *  2048-row leaf), /NCOPIES with. Rows are read as 16B uint4 chunks. */
template <int NCOPIES, typename BIN_T = uint8_t, typename HIST_T = float>
__global__ void k_hist(const BIN_T* __restrict__ rows, int stride,
                       const uint32_t* __restrict__ idx_base,
                       const int* __restrict__ leaf_begin, const int* __restrict__ leaf_cnt,
                       const LeafStat* __restrict__ stats, const int* __restrict__ leaf_slot,
                       const int* __restrict__ leafA_ptr, const int* __restrict__ counters,
                       int leafB_from_counters, const float* __restrict__ g,
                       const float* __restrict__ h, const FeatMeta* __restrict__ fm,
                       int feat_begin, int feat_end, int part_bin_base, int part_bins,
                       HIST_T* __restrict__ hist_base, size_t slot_stride) {
  const int leafA = *leafA_ptr;
  if (leafA < 0) return;
  const int leafB = leafB_from_counters ? counters[0] - 1 : -1;
  int leaf = leafA;
  if (leafB >= 0 && stats[leafB].cnt < stats[leafA].cnt) leaf = leafB;
  const int begin = leaf_begin[leaf];
  const int cnt = leaf_cnt[leaf];
  const uint32_t* idx = idx_base + begin;
  HIST_T* ghist = hist_base + static_cast<size_t>(leaf_slot[leaf]) * slot_stride +
                  static_cast<size_t>(part_bin_base) * 2;

  // LDS layout [bin][NCOPIES][g,h] with a non-power-of-2 bin stride (2*NCOPIES+2):
  // copies of one bin sit in adjacent banks (same-bin leaves drain through NCOPIES
  // banks in parallel) and consecutive bins rotate across all 32 banks.
  // HIST_T=double is the gpu_use_dp parity mode (fp64 accumulation end to end).
  constexpr int kBinStride = 2 * NCOPIES + 2;
  extern __shared__ char k_hist_smem[];
  HIST_T* lh = reinterpret_cast<HIST_T*>(k_hist_smem);  // part_bins * kBinStride
  __shared__ int loff[256];
  const int nfeat = feat_end - feat_begin;
  for (int i = threadIdx.x; i < nfeat; i += blockDim.x)
    loff[i] = fm[feat_begin + i].bin_off - part_bin_base;
  const int nelem = part_bins * kBinStride;
  for (int i = threadIdx.x; i < nelem; i += blockDim.x) lh[i] = HIST_T(0);
  __syncthreads();

  const int my_copy = (threadIdx.x % NCOPIES) * 2;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  const int nthreads = blockDim.x * gridDim.x;
  const int c0 = feat_begin & ~15;
  for (int i = tid; i < cnt; i += nthreads) {
    const uint32_t r = idx[i];
    const float gi = g[r];
    const float hi = h[r];
    const BIN_T* rp = rows + static_cast<size_t>(r) * stride;
    if constexpr (sizeof(BIN_T) == 1) {
      // uint8 rows: 16B uint4 vector loads (the headline path)
      for (int c = c0; c < feat_end; c += 16) {
        const uint4 v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const uint8_t*>(rp) + c);
        const uint32_t w[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int f = c + j;
          if (f < feat_begin || f >= feat_end) continue;
          const int b = (w[j >> 2] >> ((j & 3) * 8)) & 0xFF;
          HIST_T* dst = lh + (loff[f - feat_begin] + b) * kBinStride + my_copy;
          atomicAdd(dst, HIST_T(gi));
          atomicAdd(dst + 1, HIST_T(hi));
        }
      }
    } else {
      // uint16 rows (max_bin > 256): plain element loads — capability path,
      // parity: reference CUDAConstructHistogramDenseKernel 16-bit bins
      for (int f = feat_begin; f < feat_end; ++f) {
        const int b = rp[f];
        HIST_T* dst = lh + (loff[f - feat_begin] + b) * kBinStride + my_copy;
        atomicAdd(dst, HIST_T(gi));
        atomicAdd(dst + 1, HIST_T(hi));
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < part_bins * 2; i += blockDim.x) {
    const int bin = i >> 1;
    const int gh = i & 1;
    HIST_T v = HIST_T(0);
#pragma unroll
    for (int cpy = 0; cpy < NCOPIES; ++cpy) v += lh[bin * kBinStride + cpy * 2 + gh];
    if (v != HIST_T(0)) atomicAdd(&ghist[i], v);
  }
}

template <typename HIST_T = float>
__global__ void k_hist_zero(HIST_T* hist_base, size_t slot_stride,
                            const int* __restrict__ counters, int slot_from_counters,
                            int literal_slot, const int* __restrict__ leafA_ptr, int n) {
  if (*leafA_ptr < 0) return;
  const int slot = slot_from_counters ? counters[0] - 1 : literal_slot;
  HIST_T* hist = hist_base + static_cast<size_t>(slot) * slot_stride;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < n; i += blockDim.x * gridDim.x) hist[i] = HIST_T(0);
}

/*! larger-child histogram = parent (in place at the larger leaf's slot) - smaller. */
template <typename HIST_T = float>
__global__ void k_hist_subtract(HIST_T* __restrict__ hist_base, size_t slot_stride,
                                const int* __restrict__ leaf_slot,
                                const LeafStat* __restrict__ stats,
                                const int* __restrict__ Lptr,
                                const int* __restrict__ counters, int n) {
  const int L = *Lptr;
  if (L < 0) return;
  const int R = counters[0] - 1;
  const int smaller = stats[L].cnt <= stats[R].cnt ? L : R;
  const int larger = smaller == L ? R : L;
  HIST_T* big = hist_base + static_cast<size_t>(leaf_slot[larger]) * slot_stride;
  const HIST_T* small = hist_base + static_cast<size_t>(leaf_slot[smaller]) * slot_stride;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < n; i += blockDim.x * gridDim.x) big[i] -= small[i];
}

// ------------------------------------------------------------------ root setup
__global__ void k_init_root(int* leaf_begin, int* leaf_cnt, int* leaf_slot, LeafStat* stats,
                            int used_cnt, int64_t* gbuf, int* counters, int* root_leaf,
                            int* minus1, double* leaf_bounds,
                            unsigned long long* leaf_branch) {
  leaf_bounds[0] = -1e308;  // monotone output bounds of the root leaf
  leaf_bounds[1] = 1e308;
  leaf_branch[0] = leaf_branch[1] = leaf_branch[2] = leaf_branch[3] = 0ull;
  leaf_begin[0] = 0;
  leaf_cnt[0] = used_cnt;
  leaf_slot[0] = 0;
  stats[0].sum_g = 0.0;
  stats[0].sum_h = 0.0;
  stats[0].cnt = used_cnt;
  stats[0].depth = 0;
  gbuf[0] = used_cnt;
  counters[0] = 1;  // num_leaves
  counters[1] = 0;  // split log index
  *root_leaf = 0;
  *minus1 = -1;
}

__global__ void k_root_sums(const uint32_t* __restrict__ idx, int cnt,
                            const float* __restrict__ g, const float* __restrict__ h,
                            LeafStat* stat) {
  __shared__ double sg[4], sh[4];
  double tg = 0, th = 0;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < cnt; i += blockDim.x * gridDim.x) {
    const uint32_t r = idx[i];
    tg += g[r];
    th += h[r];
  }
  for (int d = 32; d > 0; d >>= 1) {
    tg += __shfl_down(tg, d);
    th += __shfl_down(th, d);
  }
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    sg[wave] = tg;
    sh[wave] = th;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double bg = 0, bh = 0;
    for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) {
      bg += sg[w];
      bh += sh[w];
    }
    atomicAdd(&stat->sum_g, bg);
    atomicAdd(&stat->sum_h, bh);
  }
}

__global__ void k_set_root_global_cnt(LeafStat* stats, const int64_t* gbuf,
                                      GainParams p) {
  stats[0].cnt = static_cast<int>(gbuf[0]);
  stats[0].parent_out = d_leaf_out(stats[0].sum_g, stats[0].sum_h, p);
}

// ------------------------------------------------------------------ best split
/*! one wave per (feature, child); blockIdx.y selects leafA/leafB. */
template <typename HIST_T = float>
__global__ void __launch_bounds__(64) k_best_feat(
    const HIST_T* __restrict__ hist_base, size_t slot_stride, const int* __restrict__ leaf_slot,
    const FeatMeta* __restrict__ fm, int nf, const LeafStat* __restrict__ stats,
    const int* __restrict__ leafA_ptr, const int* __restrict__ counters,
    int leafB_from_counters, GainParams p, const int8_t* __restrict__ feat_mask,
    const int8_t* __restrict__ mono, const double* __restrict__ leaf_bounds,
    const unsigned long long* __restrict__ group_masks,
    const unsigned long long* __restrict__ leaf_branch,
    const float* __restrict__ cegb_coupled, const float* __restrict__ cegb_lazy,
    SplitRec* __restrict__ out) {
  const int f = blockIdx.x;
  const int which = blockIdx.y;
  const int leafA = *leafA_ptr;
  const int leafB = leafB_from_counters ? counters[0] - 1 : -1;
  const int leaf = which == 0 ? leafA : leafB;
  if (f >= nf || leaf < 0) return;
  const int lane = threadIdx.x;
  SplitRec& rec = out[which * nf + f];
  const FeatMeta m = fm[f];
  if (lane == 0) {
    rec.valid = 0;
    rec.feature = f;
    rec.cat_mask[0] = rec.cat_mask[1] = rec.cat_mask[2] = rec.cat_mask[3] = 0;
  }
  if (f < p.own_fb || f >= p.own_fe) return;  // another rank owns this feature's scan
  if (feat_mask != nullptr && !feat_mask[f]) return;
  if (p.n_interaction_groups > 0) {
    // interaction constraints (4-word inner-feature bitmasks, nf <= 256):
    // feature f is allowed iff some group covers the leaf's branch features + f
    const int fw = f >> 6;
    bool ok = false;
    for (int g = 0; g < p.n_interaction_groups; ++g) {
      const unsigned long long* gm = group_masks + 4 * g;
      bool cover = true;
      for (int w = 0; w < 4; ++w) {
        const unsigned long long need =
            leaf_branch[4 * leaf + w] | (w == fw ? (1ull << (f & 63)) : 0ull);
        cover = cover && (need & ~gm[w]) == 0ull;
      }
      ok = ok || cover;
    }
    if (!ok) return;
  }
  const int split_idx = counters[1];
  if (p.bynode_frac < 1.0f) {
    // deterministic per-(split, child, feature) Bernoulli mask; one hashed anchor
    // feature per node is always kept so a node can never lose every feature
    const int anchor = static_cast<int>(
        d_hash3(p.rng_seed, static_cast<uint32_t>(split_idx * 2 + which), 0xFFFFu) %
        static_cast<uint32_t>(nf));
    const uint32_t hv = d_hash3(p.rng_seed, static_cast<uint32_t>(split_idx * 2 + which),
                                static_cast<uint32_t>(f));
    if (f != anchor && (hv & 0xFFFFFF) * (1.0f / 16777216.0f) >= p.bynode_frac) return;
  }
  const LeafStat st = stats[leaf];
  const double sum_g = st.sum_g;
  const double sum_h = st.sum_h;
  const int num_data = st.cnt;
  if (num_data < 2 * p.min_data) return;
  if (p.max_depth > 0 && st.depth >= p.max_depth) return;
  const double cnt_factor = (num_data > 0 && sum_h > 0) ? num_data / sum_h : 1.0;
  const double leaf_parent_out = st.parent_out;
  const double parent_gain = d_leaf_gain_sm(sum_g, sum_h, num_data, leaf_parent_out, p);
  const double min_gain_shift = parent_gain + p.min_gain_to_split;

  const HIST_T* fh = hist_base + static_cast<size_t>(leaf_slot[leaf]) * slot_stride +
                     static_cast<size_t>(m.bin_off) * 2;
  double g_nan = 0, h_nan = 0;
  const bool has_nan = m.nan_bin >= 0;
  if (has_nan) {
    g_nan = fh[2 * m.nan_bin];
    h_nan = fh[2 * m.nan_bin + 1];
  }

  double best_gain = -1e308;
  int best_bin = -1, best_dl = 0;
  double best_lg = 0, best_lh = 0;
  // monotone constraints: per-leaf output bounds (BasicLeafConstraints, same
  // clamp-then-score semantics as the CPU oracle) + per-feature direction
  const int8_t mc = mono != nullptr ? mono[f] : static_cast<int8_t>(0);
  const double blo = mono != nullptr ? leaf_bounds[2 * leaf] : -1e308;
  const double bhi = mono != nullptr ? leaf_bounds[2 * leaf + 1] : 1e308;

  if (m.is_cat) {
    const double l2c = p.l2 + p.cat_l2;
    if (m.num_bin > p.max_cat_to_onehot && m.num_bin <= 64) {
      // sorted-subset scan (CPU-oracle parity: FindBestThresholdCategorical).
      // One wave: lane = bin; bitonic sort by grad/(hess+cat_smooth); prefix
      // scans give the k-subset sums from both ends; the winner's bin subset is
      // reconstructed as a 64-bit mask carried in SplitRec.cat_mask.
      int orig = lane;
      double gb = 0.0, hb = 0.0, ratio = 1e300;
      bool elig = false;
      if (lane < m.num_bin) {
        gb = fh[2 * lane];
        hb = fh[2 * lane + 1];
        elig = hb * cnt_factor >= p.cat_smooth;
        ratio = elig ? gb / (hb + p.cat_smooth) : 1e300;
      }
      // wave64 bitonic sort ascending by (ratio, orig)
      for (int k2 = 2; k2 <= 64; k2 <<= 1) {
        for (int j = k2 >> 1; j > 0; j >>= 1) {
          const double r2 = __shfl_xor(ratio, j);
          const double g2 = __shfl_xor(gb, j);
          const double h2 = __shfl_xor(hb, j);
          const int b2 = __shfl_xor(orig, j);
          const bool lower = (lane & j) == 0;
          const bool asc = (lane & k2) == 0;
          const bool other_lt = r2 < ratio || (r2 == ratio && b2 < orig);
          const bool take = lower ? (other_lt == asc) : (other_lt != asc);
          if (take) {
            ratio = r2;
            gb = g2;
            hb = h2;
            orig = b2;
          }
        }
      }
      const uint64_t elig_ballot = __ballot(ratio < 1e300);
      const int n_elig = __popcll(elig_ballot);
      if (n_elig >= 2) {
        // inclusive prefix sums over the sorted order
        double pg = gb, ph = hb;
        for (int d = 1; d < 64; d <<= 1) {
          const double tg = __shfl_up(pg, d);
          const double th = __shfl_up(ph, d);
          if (lane >= d) {
            pg += tg;
            ph += th;
          }
        }
        const int limit = min(p.max_cat_threshold, n_elig - 1);
        const double tot_g = __shfl(pg, n_elig - 1);
        const double tot_h = __shfl(ph, n_elig - 1);
        double bg_cat = -1e308;
        int b_dir = 0;
        double b_lg = 0, b_lh = 0;
        // dir-1 ("from the high end") prefix lookups, hoisted so every lane
        // participates in the shuffles (no divergent-source reads)
        const int lo_idx_raw = n_elig - 2 - lane;
        const double pre_g = __shfl(pg, lo_idx_raw < 0 ? 0 : lo_idx_raw);
        const double pre_h = __shfl(ph, lo_idx_raw < 0 ? 0 : lo_idx_raw);
        // lane i evaluates subset size k = i+1 for both directions
        if (lane < limit) {
          for (int dir = 0; dir < 2; ++dir) {
            double sgl, shl;
            if (dir == 0) {
              sgl = pg;
              shl = ph;
            } else {
              if (lo_idx_raw < 0) continue;
              sgl = tot_g - pre_g;
              shl = tot_h - pre_h;
            }
            const double sgr = sum_g - sgl, shr = sum_h - shl;
            const int lc = static_cast<int>(shl * cnt_factor + 0.5);
            const int rc = num_data - lc;
            if (shl < p.min_hess || lc < p.min_data) continue;
            if (shr < p.min_hess || rc < p.min_data) continue;
            const double gain = d_split_gain_l2(sgl, shl, sgr, shr, p, l2c);
            if (gain <= min_gain_shift) continue;
            if (gain > bg_cat) {
              bg_cat = gain;
              b_dir = dir;
              b_lg = sgl;
              b_lh = shl;
            }
          }
        }
        // wave argmax over lanes (tie: smaller subset wins)
        int b_k = lane;
        for (int d = 32; d > 0; d >>= 1) {
          const double og = __shfl_xor(bg_cat, d);
          const int ok = __shfl_xor(b_k, d);
          const int od = __shfl_xor(b_dir, d);
          const double olg = __shfl_xor(b_lg, d);
          const double olh = __shfl_xor(b_lh, d);
          if (og > bg_cat || (og == bg_cat && ok < b_k)) {
            bg_cat = og;
            b_k = ok;
            b_dir = od;
            b_lg = olg;
            b_lh = olh;
          }
        }
        if (bg_cat > -1e307) {
          // membership mask over ORIGINAL bins for the winning subset
          const bool member = b_dir == 0 ? (lane <= b_k)
                                         : (lane >= n_elig - 1 - b_k && lane < n_elig);
          uint64_t mask = member && lane < n_elig ? (1ull << orig) : 0ull;
          for (int d = 32; d > 0; d >>= 1) mask |= __shfl_xor(mask, d);
          if (lane == 0) {
            // bin 0 (NaN/unseen) must route RIGHT: the raw-value predictor
            // cannot express the dummy category -1 in a left bitset, so
            // complement the (symmetric) subset instead (CPU-oracle parity)
            if (mask & 1ull) {
              const unsigned long long all =
                  m.num_bin >= 64 ? ~0ull : ((1ull << m.num_bin) - 1ull);
              mask = ~mask & all;
              b_lg = sum_g - b_lg;
              b_lh = sum_h - b_lh;
            }
            rec.valid = 1;
            rec.gain = bg_cat - min_gain_shift + p.min_gain_to_split;
            rec.feature = f;
            rec.bin = __popcll(mask);  // #cats on the left (CPU threshold semantics)
            rec.default_left = 0;
            rec.cat_mask[0] = mask;
            rec.cat_mask[1] = rec.cat_mask[2] = rec.cat_mask[3] = 0;
            rec.left_g = b_lg;
            rec.left_h = b_lh;
            rec.left_out = d_leaf_out_l2(b_lg, b_lh, p, l2c);
            rec.right_out = d_leaf_out_l2(sum_g - b_lg, sum_h - b_lh, p, l2c);
            rec.left_cnt = static_cast<int>(b_lh * cnt_factor + 0.5);
            rec.right_cnt = num_data - rec.left_cnt;
            d_apply_cegb(rec, f, num_data, p, cegb_coupled, cegb_lazy);
          }
        }
      }
      return;
    }
    if (m.num_bin > p.max_cat_to_onehot && m.num_bin <= 256) {
      // 65..256-bin sorted-subset scan: LDS bitonic sort (4 elems/lane) +
      // serial prefix, winner mask carried as 4x64-bit words in SplitRec.
      // Same math as the <=64 wave path (CPU-oracle parity).
      __shared__ double s_ratio[256], s_cg[256], s_ch[256];
      __shared__ short s_sorted_orig[256];
      __shared__ double s_pg[256], s_ph[256];
      __shared__ int s_nelig;
      int E = 1;
      while (E < m.num_bin) E <<= 1;
      for (int i = lane; i < E; i += 64) {
        double gb = 0, hb = 0, ratio = 1e300;
        if (i < m.num_bin) {
          gb = fh[2 * i];
          hb = fh[2 * i + 1];
          const bool elig = hb * cnt_factor >= p.cat_smooth;
          ratio = elig ? gb / (hb + p.cat_smooth) : 1e300;
        }
        s_ratio[i] = ratio;
        s_cg[i] = gb;
        s_ch[i] = hb;
        s_sorted_orig[i] = static_cast<short>(i);
      }
      __syncthreads();
      for (int ksz = 2; ksz <= E; ksz <<= 1) {
        for (int j = ksz >> 1; j > 0; j >>= 1) {
          for (int i = lane; i < E; i += 64) {
            const int ixj = i ^ j;
            if (ixj > i) {
              const bool asc = (i & ksz) == 0;
              const double r1 = s_ratio[i], r2 = s_ratio[ixj];
              const short o1 = s_sorted_orig[i], o2 = s_sorted_orig[ixj];
              const bool other_lt = r2 < r1 || (r2 == r1 && o2 < o1);
              if (other_lt == asc) {
                s_ratio[i] = r2;
                s_ratio[ixj] = r1;
                s_sorted_orig[i] = o2;
                s_sorted_orig[ixj] = o1;
                const double tg = s_cg[i], th = s_ch[i];
                s_cg[i] = s_cg[ixj];
                s_ch[i] = s_ch[ixj];
                s_cg[ixj] = tg;
                s_ch[ixj] = th;
              }
            }
          }
          __syncthreads();
        }
      }
      {
        int cnt_e = 0;
        for (int i = lane; i < E; i += 64)
          if (s_ratio[i] < 1e300) ++cnt_e;
        for (int d = 32; d > 0; d >>= 1) cnt_e += __shfl_down(cnt_e, d);
        if (lane == 0) s_nelig = cnt_e;
      }
      __syncthreads();
      const int n_elig = s_nelig;
      if (n_elig >= 2) {
        if (lane == 0) {
          double pg = 0, ph = 0;
          for (int i = 0; i < n_elig; ++i) {
            pg += s_cg[i];
            ph += s_ch[i];
            s_pg[i] = pg;
            s_ph[i] = ph;
          }
        }
        __syncthreads();
        const int limit = min(p.max_cat_threshold, n_elig - 1);
        const double tot_g = s_pg[n_elig - 1];
        const double tot_h = s_ph[n_elig - 1];
        double bg_cat = -1e308;
        int b_k = -1, b_dir = 0;
        double b_lg = 0, b_lh = 0;
        for (int k2 = lane; k2 < limit; k2 += 64) {
          for (int dir = 0; dir < 2; ++dir) {
            double sgl, shl;
            if (dir == 0) {
              sgl = s_pg[k2];
              shl = s_ph[k2];
            } else {
              const int idx = n_elig - 2 - k2;
              if (idx < 0) continue;
              sgl = tot_g - s_pg[idx];
              shl = tot_h - s_ph[idx];
            }
            const double sgr = sum_g - sgl, shr = sum_h - shl;
            const int lc = static_cast<int>(shl * cnt_factor + 0.5);
            const int rc = num_data - lc;
            if (shl < p.min_hess || lc < p.min_data) continue;
            if (shr < p.min_hess || rc < p.min_data) continue;
            const double gain = d_split_gain_l2(sgl, shl, sgr, shr, p, l2c);
            if (gain <= min_gain_shift) continue;
            if (gain > bg_cat || (gain == bg_cat && k2 < b_k)) {
              bg_cat = gain;
              b_k = k2;
              b_dir = dir;
              b_lg = sgl;
              b_lh = shl;
            }
          }
        }
        for (int d = 32; d > 0; d >>= 1) {
          const double og = __shfl_xor(bg_cat, d);
          const int ok = __shfl_xor(b_k, d);
          const int od = __shfl_xor(b_dir, d);
          const double olg = __shfl_xor(b_lg, d);
          const double olh = __shfl_xor(b_lh, d);
          if ((og > bg_cat) ||
              (og == bg_cat && ok >= 0 && (b_k < 0 || ok < b_k))) {
            bg_cat = og;
            b_k = ok;
            b_dir = od;
            b_lg = olg;
            b_lh = olh;
          }
        }
        if (lane == 0 && bg_cat > -1e307) {
          rec.valid = 1;
          rec.gain = bg_cat - min_gain_shift + p.min_gain_to_split;
          rec.feature = f;
          rec.default_left = 0;
          rec.cat_mask[0] = rec.cat_mask[1] = rec.cat_mask[2] = rec.cat_mask[3] = 0;
          const int lo_i = b_dir == 0 ? 0 : n_elig - 1 - b_k;
          const int hi_i = b_dir == 0 ? b_k : n_elig - 1;
          for (int i = lo_i; i <= hi_i; ++i) {
            const int ob = s_sorted_orig[i];
            rec.cat_mask[ob >> 6] |= 1ull << (ob & 63);
          }
          // bin 0 (NaN/unseen) must route RIGHT — complement the subset
          if (rec.cat_mask[0] & 1ull) {
            for (int w2 = 0; w2 < 4; ++w2) {
              const int lo2 = w2 * 64;
              unsigned long long all = 0ull;
              if (m.num_bin > lo2) {
                const int nb2 = m.num_bin - lo2;
                all = nb2 >= 64 ? ~0ull : ((1ull << nb2) - 1ull);
              }
              rec.cat_mask[w2] = ~rec.cat_mask[w2] & all;
            }
            b_lg = sum_g - b_lg;
            b_lh = sum_h - b_lh;
          }
          rec.bin = __popcll(rec.cat_mask[0]) + __popcll(rec.cat_mask[1]) +
                    __popcll(rec.cat_mask[2]) + __popcll(rec.cat_mask[3]);
          rec.left_g = b_lg;
          rec.left_h = b_lh;
          rec.left_out = d_leaf_out_l2(b_lg, b_lh, p, l2c);
          rec.right_out = d_leaf_out_l2(sum_g - b_lg, sum_h - b_lh, p, l2c);
          rec.left_cnt = static_cast<int>(b_lh * cnt_factor + 0.5);
          rec.right_cnt = num_data - rec.left_cnt;
          d_apply_cegb(rec, f, num_data, p, cegb_coupled, cegb_lazy);
        }
      }
      return;
    }
    for (int b = lane; b < m.num_bin; b += 64) {
      const double gl = fh[2 * b], hl = fh[2 * b + 1];
      const double gr = sum_g - gl, hr = sum_h - hl;
      const int lc = static_cast<int>(hl * cnt_factor + 0.5);
      const int rc = num_data - lc;
      if (hl < p.min_hess || lc < p.min_data || hr < p.min_hess || rc < p.min_data) continue;
      const double gain = d_split_gain_l2(gl, hl, gr, hr, p, l2c);
      if (gain <= min_gain_shift) continue;
      if (gain > best_gain || (gain == best_gain && b < best_bin)) {
        best_gain = gain;
        best_bin = b;
        best_dl = 0;
        best_lg = gl;
        best_lh = hl;
      }
    }
  } else {
    const int nb = m.num_numeric_bin;
    // with a NaN bin the last numeric bin IS a valid threshold: left = all
    // numeric values, right = missing only (all-numeric-vs-NaN split)
    const int t_max = has_nan ? nb - 1 : nb - 2;
    double carry_g = 0, carry_h = 0;
    for (int chunk = 0; chunk * 64 < nb; ++chunk) {
      const int b = chunk * 64 + lane;
      double gb = b < nb ? static_cast<double>(fh[2 * b]) : 0.0;
      double hb = b < nb ? static_cast<double>(fh[2 * b + 1]) : 0.0;
      for (int d = 1; d < 64; d <<= 1) {
        const double tg = __shfl_up(gb, d);
        const double th = __shfl_up(hb, d);
        if (lane >= d) {
          gb += tg;
          hb += th;
        }
      }
      const double GL = carry_g + gb;
      const double HL = carry_h + hb;
      carry_g += __shfl(gb, 63);
      carry_h += __shfl(hb, 63);
      if (b > t_max) continue;
      if (p.extra_trees) {
        // extra_trees: only the hashed random threshold of this feature is eligible
        const int rand_t = static_cast<int>(
            d_hash3(p.rng_seed ^ 0xA5A5A5A5u, static_cast<uint32_t>(split_idx * 2 + which),
                    static_cast<uint32_t>(f)) % static_cast<uint32_t>(t_max + 1));
        if (b != rand_t) continue;
      }
      const int n_var = has_nan ? 2 : 1;
      for (int v = 0; v < n_var; ++v) {
        const bool ml = v == 1;
        const double gl = GL + (ml ? g_nan : 0.0);
        const double hl = HL + (ml ? h_nan : 0.0);
        const double gr = sum_g - gl, hr = sum_h - hl;
        const int lc = static_cast<int>(hl * cnt_factor + 0.5);
        const int rc = num_data - lc;
        if (hl < p.min_hess || lc < p.min_data) continue;
        if (hr < p.min_hess || rc < p.min_data) continue;
        double lo = d_smooth_out(d_leaf_out(gl, hl, p), lc, leaf_parent_out, p);
        double ro = d_smooth_out(d_leaf_out(gr, hr, p), rc, leaf_parent_out, p);
        lo = fmin(fmax(lo, blo), bhi);
        ro = fmin(fmax(ro, blo), bhi);
        if (mc > 0 && lo > ro) continue;
        if (mc < 0 && lo < ro) continue;
        const double gain = d_gain_out(gl, hl, lo, p) + d_gain_out(gr, hr, ro, p);
        if (gain <= min_gain_shift) continue;
        if (gain > best_gain || (gain == best_gain && b < best_bin)) {
          best_gain = gain;
          best_bin = b;
          best_dl = ml ? 1 : 0;
          best_lg = gl;
          best_lh = hl;
        }
      }
    }
  }
  for (int d = 32; d > 0; d >>= 1) {
    const double og = __shfl_xor(best_gain, d);
    const int ob = __shfl_xor(best_bin, d);
    const int odl = __shfl_xor(best_dl, d);
    const double olg = __shfl_xor(best_lg, d);
    const double olh = __shfl_xor(best_lh, d);
    const bool take = (og > best_gain) ||
                      (og == best_gain && ob >= 0 && (best_bin < 0 || ob < best_bin));
    if (take) {
      best_gain = og;
      best_bin = ob;
      best_dl = odl;
      best_lg = olg;
      best_lh = olh;
    }
  }
  if (lane == 0 && best_bin >= 0) {
    rec.valid = 1;
    rec.gain = best_gain - min_gain_shift + p.min_gain_to_split;
    if (mc != 0 && p.mono_penalty > 0.0) {
      // depth-decaying multiplicative penalty on monotone splits (CPU-oracle
      // ComputeMonotoneSplitGainPenalty parity)
      const double pen = p.mono_penalty;
      const int dep = st.depth;
      double factor;
      if (pen >= dep + 1.0) factor = 1e-15;
      else if (pen <= 1.0) factor = 1.0 - pen / exp2((double)dep) + 1e-15;
      else factor = 1.0 - exp2(pen - 1.0 - dep) + 1e-15;
      rec.gain *= factor;
    }
    rec.feature = f;
    rec.bin = best_bin;
    rec.default_left = best_dl;
    if (m.is_cat && best_bin == 0) {
      // one-hot winner on the NaN/unseen bin: missing must route RIGHT at
      // predict time, so emit the complement subset ("every bin except 0")
      for (int w2 = 0; w2 < 4; ++w2) {
        const int lo2 = w2 * 64;
        unsigned long long all = 0ull;
        if (m.num_bin > lo2) {
          const int nb2 = m.num_bin - lo2;
          all = nb2 >= 64 ? ~0ull : ((1ull << nb2) - 1ull);
        }
        rec.cat_mask[w2] = all;
      }
      rec.cat_mask[0] &= ~1ull;
      best_lg = sum_g - best_lg;
      best_lh = sum_h - best_lh;
      rec.bin = m.num_bin - 1;
    }
    rec.left_g = best_lg;
    rec.left_h = best_lh;
    const double rg = sum_g - best_lg, rh = sum_h - best_lh;
    if (m.is_cat) {
      rec.left_out = d_leaf_out(best_lg, best_lh, p);
      rec.right_out = d_leaf_out(rg, rh, p);
    } else {
      const int w_lc = static_cast<int>(best_lh * cnt_factor + 0.5);
      rec.left_out = fmin(fmax(d_smooth_out(d_leaf_out(best_lg, best_lh, p), w_lc,
                                            leaf_parent_out, p), blo), bhi);
      rec.right_out = fmin(fmax(d_smooth_out(d_leaf_out(rg, rh, p), num_data - w_lc,
                                             leaf_parent_out, p), blo), bhi);
    }
    rec.left_cnt = static_cast<int>(best_lh * cnt_factor + 0.5);
    rec.right_cnt = num_data - rec.left_cnt;
    d_apply_cegb(rec, f, num_data, p, cegb_coupled, cegb_lazy);
  }
}

/*! reduce per-feature records to one per leaf; blockIdx.x = which child. */
__global__ void k_best_leaf_overall(const SplitRec* __restrict__ feat_best, int nf,
                                    SplitRec* __restrict__ leaf_best,
                                    const int* __restrict__ leafA_ptr,
                                    const int* __restrict__ counters,
                                    int leafB_from_counters,
                                    SplitRec* __restrict__ winner,
                                    int* __restrict__ winner_leaf) {
  // Fused per-leaf argmax (both fresh children) + overall winner selection.
  // One block: the per-leaf reduction is tiny (nf recs), and fusing removes two
  // kernel launches per split from the device-driven loop.
  const int leafA = *leafA_ptr;
  const int leafB = leafB_from_counters ? counters[0] - 1 : -1;
  const int num_leaves = counters[0];
  __shared__ int s_idx[256];
  __shared__ double s_gain[256];
  const int tid = threadIdx.x;
  for (int which = 0; which < 2; ++which) {
    const int leaf = which == 0 ? leafA : leafB;
    if (leaf < 0) continue;
    const SplitRec* cand = feat_best + which * nf;
    int bi = -1;
    double bg = -1e308;
    for (int f = tid; f < nf; f += blockDim.x) {
      if (cand[f].valid &&
          (cand[f].gain > bg || (cand[f].gain == bg && (bi < 0 || f < bi)))) {
        bg = cand[f].gain;
        bi = f;
      }
    }
    s_idx[tid] = bi;
    s_gain[tid] = bg;
    __syncthreads();
    for (int r = blockDim.x / 2; r > 0; r >>= 1) {
      if (tid < r) {
        if (s_idx[tid + r] >= 0 &&
            (s_idx[tid] < 0 || s_gain[tid + r] > s_gain[tid] ||
             (s_gain[tid + r] == s_gain[tid] && s_idx[tid + r] < s_idx[tid]))) {
          s_idx[tid] = s_idx[tid + r];
          s_gain[tid] = s_gain[tid + r];
        }
      }
      __syncthreads();
    }
    if (tid == 0) {
      if (s_idx[0] >= 0) leaf_best[leaf] = cand[s_idx[0]];
      else {
        leaf_best[leaf].valid = 0;
        leaf_best[leaf].gain = -1e308;
      }
    }
    __syncthreads();
  }
  // overall winner over every live leaf (requires strictly positive improvement)
  int bi = -1;
  double bg = 0.0;
  for (int l = tid; l < num_leaves; l += blockDim.x) {
    if (leaf_best[l].valid && leaf_best[l].gain > bg) {
      bg = leaf_best[l].gain;
      bi = l;
    }
  }
  s_idx[tid] = bi;
  s_gain[tid] = bg;
  __syncthreads();
  for (int r = blockDim.x / 2; r > 0; r >>= 1) {
    if (tid < r) {
      if (s_idx[tid + r] >= 0 &&
          (s_idx[tid] < 0 || s_gain[tid + r] > s_gain[tid] ||
           (s_gain[tid + r] == s_gain[tid] && s_idx[tid + r] < s_idx[tid]))) {
        s_idx[tid] = s_idx[tid + r];
        s_gain[tid] = s_gain[tid + r];
      }
    }
    __syncthreads();
  }
  if (tid == 0) {
    *winner_leaf = s_idx[0];
    if (s_idx[0] >= 0) *winner = leaf_best[s_idx[0]];
  }
}

// ------------------------------------------------------------------ partition
/*! Deterministic block-scan partition of leaf L (grid-stride, chunked by 256-row
 *  blocks round-robined over the grid). Three phases: mark+count per block, block
 *  offset scan, ranked scatter. No same-address global atomics. */
__device__ __forceinline__ int part_decide(int b, int thr_bin, int nan_bin, int default_left,
                                           int is_cat, unsigned long long m0,
                                           unsigned long long m1, unsigned long long m2,
                                           unsigned long long m3) {
  if (is_cat) {
    if ((m0 | m1 | m2 | m3) != 0ull) {  // sorted-subset split (<=256 bins)
      const unsigned long long w = b < 64 ? m0 : b < 128 ? m1 : b < 192 ? m2 : m3;
      return (w >> (b & 63)) & 1ull;
    }
    return b == thr_bin ? 1 : 0;  // one-hot split
  }
  if (nan_bin >= 0 && b == nan_bin) return default_left;
  return b <= thr_bin ? 1 : 0;
}

template <typename BIN_T = uint8_t>
__global__ void k_part_mark(const uint32_t* __restrict__ idx_base,
                            const int* __restrict__ leaf_begin,
                            const int* __restrict__ leaf_cnt,
                            const int* __restrict__ Lptr, const SplitRec* __restrict__ win,
                            const FeatMeta* __restrict__ fm, const BIN_T* __restrict__ cols,
                            int num_data, uint8_t* __restrict__ marks,
                            int* __restrict__ block_cnt) {
  __shared__ int s_cnt[4];
  const int L = *Lptr;
  if (L < 0) return;
  const int f = win->feature;
  const FeatMeta m = fm[f];
  const BIN_T* colbins2 = cols + static_cast<size_t>(f) * num_data;
  const int thr_bin = win->bin;
  const int nan_bin = m.is_cat ? -1 : m.nan_bin;
  const int default_left = win->default_left;
  const int cat_onehot = m.is_cat;
  const unsigned long long cm0 = win->cat_mask[0], cm1 = win->cat_mask[1];
  const unsigned long long cm2 = win->cat_mask[2], cm3 = win->cat_mask[3];
  const int begin = leaf_begin[L];
  const int cnt = leaf_cnt[L];
  const uint32_t* idx = idx_base + begin;
  const int chunk_stride = gridDim.x * blockDim.x;
  int local = 0;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < cnt; i += chunk_stride) {
    const int b = colbins2[idx[i]];
    const int go =
        part_decide(b, thr_bin, nan_bin, default_left, cat_onehot, cm0, cm1, cm2, cm3);
    marks[i] = static_cast<uint8_t>(go);
    local += go;
  }
  for (int d = 32; d > 0; d >>= 1) local += __shfl_down(local, d);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) s_cnt[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    int c = 0;
    for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) c += s_cnt[w];
    block_cnt[blockIdx.x] = c;
  }
}

/*! parallel exclusive scan over per-block left counts and own-row counts (closed form
 *  for the grid-stride chunk decomposition). l_off[b] = left write base; r_off[b] holds
 *  own_prefix - left_prefix (scatter adds total_left from ctr); ctr[0] = total_left. */
__global__ void k_part_scan(const int* __restrict__ block_cnt, int nblocks,
                            const int* __restrict__ leaf_cnt,
                            const int* __restrict__ Lptr,
                            int* __restrict__ l_off, int* __restrict__ r_off,
                            int* __restrict__ ctr) {
  const int L = *Lptr;
  if (L < 0) return;
  __shared__ int s_wl[4], s_wo[4];
  __shared__ int carry_l, carry_o;
  if (threadIdx.x == 0) {
    carry_l = 0;
    carry_o = 0;
  }
  __syncthreads();
  const int cnt = leaf_cnt[L];
  const int bs = 256;
  const int nchunks = (cnt + bs - 1) / bs;
  const int q = nblocks > 0 ? nchunks / nblocks : 0;
  const int r = nblocks > 0 ? nchunks % nblocks : 0;
  const int last_b = nchunks > 0 ? (nchunks - 1) % nblocks : 0;
  const int last_sz = nchunks > 0 ? cnt - (nchunks - 1) * bs : 0;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x / 64;
  for (int base = 0; base < nblocks; base += blockDim.x) {
    const int b = base + static_cast<int>(threadIdx.x);
    int vl = b < nblocks ? block_cnt[b] : 0;
    int vo = 0;
    if (b < nblocks) {
      vo = (q + (b < r ? 1 : 0)) * bs;
      if (nchunks > 0 && b == last_b) vo -= bs - last_sz;
    }
    int il = vl, io = vo;
    for (int d = 1; d < 64; d <<= 1) {
      const int tl = __shfl_up(il, d);
      const int to = __shfl_up(io, d);
      if (lane >= d) {
        il += tl;
        io += to;
      }
    }
    if (lane == 63) {
      s_wl[wave] = il;
      s_wo[wave] = io;
    }
    __syncthreads();
    int wbl = 0, wbo = 0;
    for (int w = 0; w < wave; ++w) {
      wbl += s_wl[w];
      wbo += s_wo[w];
    }
    const int excl_l = carry_l + wbl + il - vl;
    const int excl_o = carry_o + wbo + io - vo;
    if (b < nblocks) {
      l_off[b] = excl_l;
      r_off[b] = excl_o - excl_l;
    }
    __syncthreads();
    if (threadIdx.x == blockDim.x - 1) {
      carry_l += wbl + il;
      carry_o += wbo + io;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) ctr[0] = carry_l;
}

__global__ void k_part_scatter(const uint32_t* __restrict__ idx_base,
                               uint32_t* __restrict__ tmp_base,
                               const int* __restrict__ leaf_begin,
                               const int* __restrict__ leaf_cnt,
                               const int* __restrict__ Lptr,
                               const uint8_t* __restrict__ marks,
                               const int* __restrict__ l_off, const int* __restrict__ r_off,
                               const int* __restrict__ ctr) {
  const int L = *Lptr;
  if (L < 0) return;
  __shared__ int s_l[4], s_n[4];
  __shared__ int s_lbase, s_rbase;
  const int begin = leaf_begin[L];
  const int cnt = leaf_cnt[L];
  const uint32_t* idx = idx_base + begin;
  uint32_t* tmp = tmp_base + begin;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x / 64;
  const uint64_t lt_mask = (1ull << lane) - 1;
  if (threadIdx.x == 0) {
    s_lbase = l_off[blockIdx.x];
    s_rbase = ctr[0] + r_off[blockIdx.x];
  }
  __syncthreads();
  const int chunk_stride = gridDim.x * blockDim.x;
  // block-uniform loop bound: __syncthreads below requires all waves to iterate together
  for (int base = blockIdx.x * blockDim.x; base < cnt; base += chunk_stride) {
    const int i0 = base + static_cast<int>(threadIdx.x);
    const bool active = i0 < cnt;
    int go = 0;
    uint32_t rv = 0;
    if (active) {
      rv = idx[i0];
      go = marks[i0];
    }
    const uint64_t bl = __ballot(active && go);
    const uint64_t bn = __ballot(active);
    const int lrank = __popcll(bl & lt_mask);
    const int nrank = __popcll(bn & lt_mask);
    if (lane == 0) {
      s_l[wave] = __popcll(bl);
      s_n[wave] = __popcll(bn);
    }
    __syncthreads();
    int wl = 0, wn = 0;
    for (int w = 0; w < wave; ++w) {
      wl += s_l[w];
      wn += s_n[w];
    }
    int tot_l = wl, tot_n = wn;
    for (int w = wave; w < static_cast<int>(blockDim.x / 64); ++w) {
      tot_l += s_l[w];
      tot_n += s_n[w];
    }
    if (active) {
      if (go) tmp[s_lbase + wl + lrank] = rv;
      else tmp[s_rbase + (wn + nrank) - (wl + lrank)] = rv;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      s_lbase += tot_l;
      s_rbase += tot_n - tot_l;
    }
    __syncthreads();
  }
}

/*! Whole partition in ONE cooperative launch: mark -> block-offset scan ->
 *  ranked scatter -> copy-back, separated by grid.sync() instead of kernel
 *  boundaries. Saves three dispatches per split; bit-identical to the 4-kernel
 *  path (same per-phase code). All blocks reach every grid.sync even when the
 *  split is a no-op (L < 0). */
__global__ void k_part_fused(const uint32_t* __restrict__ idx_base,
                             uint32_t* __restrict__ tmp_base,
                             const int* __restrict__ leaf_begin,
                             const int* __restrict__ leaf_cnt,
                             const int* __restrict__ Lptr,
                             const SplitRec* __restrict__ win,
                             const FeatMeta* __restrict__ fm,
                             const uint8_t* __restrict__ cols, int num_data,
                             uint8_t* __restrict__ marks, int* __restrict__ block_cnt,
                             int* __restrict__ l_off, int* __restrict__ r_off,
                             int* __restrict__ ctr, int64_t* gbuf,
                             uint32_t* __restrict__ idx_mut) {
  const auto grid = cooperative_groups::this_grid();
  __shared__ int s_cnt[4];
  __shared__ int s_wl[4], s_wo[4];
  __shared__ int carry_l, carry_o;
  __shared__ int s_l[4], s_n[4];
  __shared__ int s_lbase, s_rbase;
  const int L = *Lptr;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x / 64;
  const int chunk_stride = gridDim.x * blockDim.x;
  int begin = 0, cnt = 0;
  if (L >= 0) {
    begin = leaf_begin[L];
    cnt = leaf_cnt[L];
  }
  // ---- phase 1: mark + per-block left counts
  if (L >= 0) {
    const int f = win->feature;
    const FeatMeta m = fm[f];
    const uint8_t* colbins2 = cols + static_cast<size_t>(f) * num_data;
    const int thr_bin = win->bin;
    const int nan_bin = m.is_cat ? -1 : m.nan_bin;
    const int default_left = win->default_left;
    const int cat_onehot = m.is_cat;
    const unsigned long long cm0 = win->cat_mask[0], cm1 = win->cat_mask[1];
    const unsigned long long cm2 = win->cat_mask[2], cm3 = win->cat_mask[3];
    const uint32_t* idx = idx_base + begin;
    int local = 0;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < cnt; i += chunk_stride) {
      const int b = colbins2[idx[i]];
      const int go =
          part_decide(b, thr_bin, nan_bin, default_left, cat_onehot, cm0, cm1, cm2, cm3);
      marks[i] = static_cast<uint8_t>(go);
      local += go;
    }
    for (int d = 32; d > 0; d >>= 1) local += __shfl_down(local, d);
    if (lane == 0) s_cnt[wave] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
      int c = 0;
      for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) c += s_cnt[w];
      block_cnt[blockIdx.x] = c;
    }
  }
  grid.sync();
  // ---- phase 2: exclusive scan of block counts (block 0 only)
  if (L >= 0 && blockIdx.x == 0) {
    if (threadIdx.x == 0) {
      carry_l = 0;
      carry_o = 0;
    }
    __syncthreads();
    const int nblocks = gridDim.x;
    const int bs = 256;
    const int nchunks = (cnt + bs - 1) / bs;
    const int q = nchunks / nblocks;
    const int r = nchunks % nblocks;
    const int last_b = nchunks > 0 ? (nchunks - 1) % nblocks : 0;
    const int last_sz = nchunks > 0 ? cnt - (nchunks - 1) * bs : 0;
    for (int base = 0; base < nblocks; base += blockDim.x) {
      const int b = base + static_cast<int>(threadIdx.x);
      int vl = b < nblocks ? block_cnt[b] : 0;
      int vo = 0;
      if (b < nblocks) {
        vo = (q + (b < r ? 1 : 0)) * bs;
        if (nchunks > 0 && b == last_b) vo -= bs - last_sz;
      }
      int il = vl, io = vo;
      for (int d = 1; d < 64; d <<= 1) {
        const int tl = __shfl_up(il, d);
        const int to = __shfl_up(io, d);
        if (lane >= d) {
          il += tl;
          io += to;
        }
      }
      if (lane == 63) {
        s_wl[wave] = il;
        s_wo[wave] = io;
      }
      __syncthreads();
      int wbl = 0, wbo = 0;
      for (int w = 0; w < wave; ++w) {
        wbl += s_wl[w];
        wbo += s_wo[w];
      }
      const int excl_l = carry_l + wbl + il - vl;
      const int excl_o = carry_o + wbo + io - vo;
      if (b < nblocks) {
        l_off[b] = excl_l;
        r_off[b] = excl_o - excl_l;
      }
      __syncthreads();
      if (threadIdx.x == blockDim.x - 1) {
        carry_l += wbl + il;
        carry_o += wbo + io;
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      ctr[0] = carry_l;
      gbuf[0] = carry_l;  // fused store-left for the (optional) RCCL sum
    }
  }
  grid.sync();
  // ---- phase 3: ranked scatter into tmp
  if (L >= 0) {
    const uint32_t* idx = idx_base + begin;
    uint32_t* tmp = tmp_base + begin;
    const uint64_t lt_mask = (1ull << lane) - 1;
    if (threadIdx.x == 0) {
      s_lbase = l_off[blockIdx.x];
      s_rbase = ctr[0] + r_off[blockIdx.x];
    }
    __syncthreads();
    for (int base = blockIdx.x * blockDim.x; base < cnt; base += chunk_stride) {
      const int i0 = base + static_cast<int>(threadIdx.x);
      const bool active = i0 < cnt;
      int go = 0;
      uint32_t rv = 0;
      if (active) {
        rv = idx[i0];
        go = marks[i0];
      }
      const uint64_t bl = __ballot(active && go);
      const uint64_t bn = __ballot(active);
      const int lrank = __popcll(bl & lt_mask);
      const int nrank = __popcll(bn & lt_mask);
      if (lane == 0) {
        s_l[wave] = __popcll(bl);
        s_n[wave] = __popcll(bn);
      }
      __syncthreads();
      int wl = 0, wn = 0;
      for (int w = 0; w < wave; ++w) {
        wl += s_l[w];
        wn += s_n[w];
      }
      int tot_l = wl, tot_n = wn;
      for (int w = wave; w < static_cast<int>(blockDim.x / 64); ++w) {
        tot_l += s_l[w];
        tot_n += s_n[w];
      }
      if (active) {
        if (go) tmp[s_lbase + wl + lrank] = rv;
        else tmp[s_rbase + (wn + nrank) - (wl + lrank)] = rv;
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        s_lbase += tot_l;
        s_rbase += tot_n - tot_l;
      }
      __syncthreads();
    }
  }
  grid.sync();
  // ---- phase 4: copy winner range back into idx
  if (L >= 0) {
    const int tid = blockIdx.x * blockDim.x + threadIdx.x;
    for (int i = tid; i < cnt; i += chunk_stride)
      idx_mut[begin + i] = tmp_base[begin + i];
  }
}

__global__ void k_copy_back(const uint32_t* __restrict__ tmp_base,
                            uint32_t* __restrict__ idx_base,
                            const int* __restrict__ leaf_begin,
                            const int* __restrict__ leaf_cnt,
                            const int* __restrict__ Lptr, const int* __restrict__ ctr,
                            int64_t* gbuf) {
  const int L = *Lptr;
  if (L < 0) return;
  // fused store-left: publish the local left count for the (optional) RCCL sum
  if (blockIdx.x == 0 && threadIdx.x == 0) gbuf[0] = ctr[0];
  const int begin = leaf_begin[L];
  const int cnt = leaf_cnt[L];
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < cnt; i += blockDim.x * gridDim.x)
    idx_base[begin + i] = tmp_base[begin + i];
}

__device__ void FinalizeBookkeeping(int* leaf_begin, int* leaf_cnt, int* leaf_slot,
                                    LeafStat* stats, const SplitRec* winner, int L,
                                    int* counters, LogEntry* log, const int* ctr,
                                    const int64_t* gbuf, const int8_t* mono,
                                    double* leaf_bounds, unsigned long long* leaf_branch);

/*! device-side split bookkeeping: segments, stats, slot map, split log. Thread 0
 *  does the bookkeeping; the whole block then zeroes the spare histogram slot the
 *  upcoming smaller-child build will accumulate into (fused k_hist_zero). */
template <typename HIST_T = float>
__global__ void k_finalize(int* leaf_begin, int* leaf_cnt, int* leaf_slot, LeafStat* stats,
                           const SplitRec* __restrict__ winner,
                           const int* __restrict__ Lptr, int* counters,
                           LogEntry* __restrict__ log, const int* __restrict__ ctr,
                           const int64_t* __restrict__ gbuf, HIST_T* hist_base,
                           size_t slot_stride, int n_elem, float* cegb_coupled,
                           const int8_t* __restrict__ mono, double* leaf_bounds,
                           unsigned long long* leaf_branch) {
  __shared__ int s_spare;
  if (threadIdx.x == 0) {
    s_spare = -1;
    const int L = *Lptr;
    if (L < 0) {
      log[counters[1]].leaf = -1;  // terminator for the host replay
    } else {
      s_spare = counters[0];
      if (cegb_coupled != nullptr) cegb_coupled[winner->feature] = 0.0f;  // feature used
      FinalizeBookkeeping(leaf_begin, leaf_cnt, leaf_slot, stats, winner, L, counters, log,
                          ctr, gbuf, mono, leaf_bounds, leaf_branch);
    }
  }
  __syncthreads();
  if (s_spare < 0) return;
  HIST_T* hist = hist_base + static_cast<size_t>(s_spare) * slot_stride;
  for (int i = threadIdx.x; i < n_elem; i += blockDim.x) hist[i] = HIST_T(0);
}

/*! single-thread bookkeeping body (called from k_finalize thread 0). */
__device__ void FinalizeBookkeeping(int* leaf_begin, int* leaf_cnt, int* leaf_slot,
                                    LeafStat* stats, const SplitRec* __restrict__ winner,
                                    int L, int* counters, LogEntry* __restrict__ log,
                                    const int* __restrict__ ctr,
                                    const int64_t* __restrict__ gbuf, const int8_t* mono,
                                    double* leaf_bounds, unsigned long long* leaf_branch) {
  const int R = counters[0];
  const int spare_slot = R;
  log[counters[1]].rec = *winner;
  log[counters[1]].leaf = L;
  counters[1] += 1;
  counters[0] += 1;
  const int left_local = ctr[0];
  const int parent_local = leaf_cnt[L];
  leaf_begin[R] = leaf_begin[L] + left_local;
  leaf_cnt[L] = left_local;
  leaf_cnt[R] = parent_local - left_local;
  const LeafStat parent = stats[L];
  const int gl = static_cast<int>(gbuf[0]);
  const int gr = parent.cnt - gl;
  const SplitRec w = *winner;
  const double right_h = parent.sum_h - w.left_h;
  const double po = (w.left_out * w.left_h + w.right_out * right_h) /
                    fmax(w.left_h + right_h, 1e-15);
  stats[L].sum_g = w.left_g;
  stats[L].sum_h = w.left_h;
  stats[L].cnt = gl;
  stats[L].parent_out = po;
  stats[L].depth = parent.depth + 1;
  stats[R].sum_g = parent.sum_g - w.left_g;
  stats[R].sum_h = right_h;
  stats[R].cnt = gr;
  stats[R].parent_out = po;
  stats[R].depth = parent.depth + 1;
  const int old_slot = leaf_slot[L];
  if (gl <= gr) {
    leaf_slot[L] = spare_slot;
    leaf_slot[R] = old_slot;
  } else {
    leaf_slot[R] = spare_slot;  // L keeps old slot
  }
  if (leaf_branch != nullptr) {
    for (int wd = 0; wd < 4; ++wd) {
      unsigned long long b = leaf_branch[4 * L + wd];
      if (wd == (w.feature >> 6)) b |= 1ull << (w.feature & 63);
      leaf_branch[4 * L + wd] = b;
      leaf_branch[4 * R + wd] = b;
    }
  }
  if (mono != nullptr) {
    // BasicLeafConstraints bound propagation (mirrors the host learner): children
    // inherit the parent bounds; a monotone split pins each side at the midpoint
    double lo = leaf_bounds[2 * L], hi = leaf_bounds[2 * L + 1];
    leaf_bounds[2 * R] = lo;
    leaf_bounds[2 * R + 1] = hi;
    const int8_t mc = mono[w.feature];
    if (mc != 0) {
      const double mid = (w.left_out + w.right_out) / 2.0;
      if (mc > 0) {
        leaf_bounds[2 * L + 1] = fmin(hi, mid);
        leaf_bounds[2 * R] = fmax(lo, mid);
      } else {
        leaf_bounds[2 * L] = fmax(lo, mid);
        leaf_bounds[2 * R + 1] = fmin(hi, mid);
      }
    }
  }
}


// -------------------------------------------------------- distributed winner sync
/*! pack this rank's overall winner (best split over its OWNED features, all
 *  leaves) into a wire record for the cross-rank allgather. */
__global__ void k_pack_winner(const SplitRec* __restrict__ winner,
                              const int* __restrict__ winner_leaf,
                              LogEntry* __restrict__ wire) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    wire->rec = *winner;
    wire->leaf = *winner_leaf;
  }
}

/*! deterministic argmax over the gathered per-rank winners; every rank picks the
 *  identical global winner (ties broken by feature index, then leaf). Replaces
 *  the reference's SyncUpGlobalBestSplit custom allreduce
 *  (parallel_tree_learner.h:209-232) with allgather + device argmax. */
__global__ void k_pick_global_winner(const LogEntry* __restrict__ wires, int world,
                                     SplitRec* __restrict__ winner,
                                     int* __restrict__ winner_leaf) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  int best = -1;
  for (int r = 0; r < world; ++r) {
    if (wires[r].leaf < 0 || !wires[r].rec.valid) continue;
    if (best < 0) {
      best = r;
      continue;
    }
    const double g1 = wires[r].rec.gain, g0 = wires[best].rec.gain;
    if (g1 > g0 ||
        (g1 == g0 && (wires[r].rec.feature < wires[best].rec.feature ||
                      (wires[r].rec.feature == wires[best].rec.feature &&
                       wires[r].leaf < wires[best].leaf)))) {
      best = r;
    }
  }
  if (best < 0) {
    *winner_leaf = -1;
  } else {
    *winner = wires[best].rec;
    *winner_leaf = wires[best].leaf;
  }
}

/*! forced split: override the winner with (leaf, feature, bin) from the
 *  forcedsplits schedule; stats from the leaf's histogram prefix (CPU
 *  MakeForcedSplit parity incl. monotone clamp/equalize). One wave. */
template <typename HIST_T = float>
__global__ void k_force_winner(const HIST_T* __restrict__ hist_base, size_t slot_stride,
                               const int* __restrict__ leaf_slot,
                               const FeatMeta* __restrict__ fm,
                               const LeafStat* __restrict__ stats, int leaf, int f,
                               int bin, GainParams p, const int8_t* __restrict__ mono,
                               const double* __restrict__ leaf_bounds,
                               SplitRec* __restrict__ winner,
                               int* __restrict__ winner_leaf) {
  const int lane = threadIdx.x;
  const FeatMeta m = fm[f];
  const HIST_T* fh = hist_base + static_cast<size_t>(leaf_slot[leaf]) * slot_stride +
                     static_cast<size_t>(m.bin_off) * 2;
  double gl = 0, hl = 0;
  for (int b = lane; b <= bin; b += 64) {
    gl += fh[2 * b];
    hl += fh[2 * b + 1];
  }
  for (int d = 32; d > 0; d >>= 1) {
    gl += __shfl_down(gl, d);
    hl += __shfl_down(hl, d);
  }
  if (lane != 0) return;
  const LeafStat st = stats[leaf];
  SplitRec rec;
  rec.valid = 1;
  rec.feature = f;
  rec.bin = bin;
  rec.default_left = 0;
  rec.cat_mask[0] = rec.cat_mask[1] = rec.cat_mask[2] = rec.cat_mask[3] = 0;
  rec.gain = 1e30;  // forced splits take precedence over gain selection
  rec.left_g = gl;
  rec.left_h = hl;
  const double cf = st.cnt > 0 && st.sum_h > 0 ? st.cnt / st.sum_h : 1.0;
  rec.left_cnt = static_cast<int>(hl * cf + 0.5);
  rec.right_cnt = st.cnt - rec.left_cnt;
  double lo = d_leaf_out(gl, hl, p);
  double ro = d_leaf_out(st.sum_g - gl, st.sum_h - hl, p);
  if (mono != nullptr) {
    const double blo = leaf_bounds[2 * leaf], bhi = leaf_bounds[2 * leaf + 1];
    lo = fmin(fmax(lo, blo), bhi);
    ro = fmin(fmax(ro, blo), bhi);
    const int8_t mc = mono[f];
    if ((mc > 0 && lo > ro) || (mc < 0 && lo < ro)) lo = ro = (lo + ro) / 2.0;
  }
  rec.left_out = lo;
  rec.right_out = ro;
  *winner = rec;
  *winner_leaf = leaf;
}

// ------------------------------------------------------------------ boosting kernels
__global__ void k_grad_binary(const double* __restrict__ score,
                              const float* __restrict__ label,
                              const float* __restrict__ weight, int n, double sigmoid,
                              double w_pos, double w_neg, float* __restrict__ g,
                              float* __restrict__ h) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double y = label[i] > 0 ? 1.0 : -1.0;
  const double lw = (y > 0 ? w_pos : w_neg) * (weight ? weight[i] : 1.0);
  const double response = -y * sigmoid / (1.0 + exp(y * sigmoid * score[i]));
  const double ar = fabs(response);
  g[i] = static_cast<float>(response * lw);
  h[i] = static_cast<float>(ar * (sigmoid - ar) * lw);
}

__global__ void k_grad_l2(const double* __restrict__ score, const float* __restrict__ label,
                          const float* __restrict__ weight, int n, float* __restrict__ g,
                          float* __restrict__ h) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double w = weight ? weight[i] : 1.0;
  g[i] = static_cast<float>(w * (score[i] - label[i]));
  h[i] = static_cast<float>(w);
}

// pointwise objective kinds (host dispatch: HIPTreeLearner::DeviceBoosting).
// Gradient formulas mirror cpp/src/objective.cpp exactly (capability parity:
// reference src/objective/cuda/cuda_regression_objective.cu, cuda_xentropy...).
enum GradKind : int {
  kGradL1 = 0,
  kGradHuber,      // a = alpha
  kGradFair,       // a = fair_c
  kGradPoisson,    // a = poisson_max_delta_step
  kGradQuantile,   // a = alpha
  kGradMape,
  kGradGamma,
  kGradTweedie,    // a = tweedie_variance_power
  kGradXent,
  kGradXentLambda,
};

__global__ void k_grad_pointwise(int kind, const double* __restrict__ score,
                                 const float* __restrict__ label,
                                 const float* __restrict__ weight, int n, double a,
                                 float* __restrict__ g, float* __restrict__ h) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double w = weight ? weight[i] : 1.0;
  const double y = label[i];
  const double s = score[i];
  const double d = s - y;
  double gv = 0.0, hv = 1.0;
  switch (kind) {
    case kGradL1:
      gv = d > 0 ? w : -w;
      hv = w;
      break;
    case kGradHuber:
      gv = fabs(d) <= a ? w * d : w * (d > 0 ? a : -a);
      hv = w;
      break;
    case kGradFair: {
      const double den = fabs(d) + a;
      gv = w * a * d / den;
      hv = w * a * a / (den * den);
      break;
    }
    case kGradPoisson: {
      const double e = exp(s);
      gv = w * (e - y);
      hv = w * exp(s + a);
      break;
    }
    case kGradQuantile:
      gv = d >= 0 ? w * (1.0 - a) : -w * a;
      hv = w;
      break;
    case kGradMape: {
      const double lw = w / fmax(1.0, fabs(y));
      gv = d > 0 ? lw : -lw;
      hv = lw;
      break;
    }
    case kGradGamma: {
      const double e = exp(-s);
      gv = w * (1.0 - y * e);
      hv = w * y * e;
      break;
    }
    case kGradTweedie: {
      const double e1 = exp((1.0 - a) * s);
      const double e2 = exp((2.0 - a) * s);
      gv = w * (-y * e1 + e2);
      hv = w * (-y * (1.0 - a) * e1 + (2.0 - a) * e2);
      break;
    }
    case kGradXent: {
      const double p = 1.0 / (1.0 + exp(-s));
      gv = w * (p - y);
      hv = w * p * (1.0 - p);
      break;
    }
    case kGradXentLambda: {
      const double epf = exp(s);
      const double hhat = log1p(epf);
      const double z = 1.0 - exp(-w * hhat);
      const double enf = 1.0 / epf;
      gv = (1.0 - y / z) * w / (1.0 + enf);
      const double c = 1.0 / (1.0 - z);
      const double dd = 1.0 + epf;
      const double aa = w * epf / (dd * dd);
      hv = aa * (1.0 + y * (1.0 - c * (1.0 + w * epf / dd * (1.0 - c))));
      break;
    }
  }
  g[i] = static_cast<float>(gv);
  h[i] = static_cast<float>(hv);
}

/*! multiclass softmax gradients, all classes at once (class-major layout,
 *  stride n). Mirrors MulticlassSoftmax::GetGradients (objective.cpp:402). */
__global__ void k_grad_multiclass(const double* __restrict__ score,
                                  const float* __restrict__ label,
                                  const float* __restrict__ weight, int n, int num_class,
                                  float* __restrict__ g, float* __restrict__ h) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double w = weight ? weight[i] : 1.0;
  const double factor = static_cast<double>(num_class) / (num_class - 1);
  double mx = -1e300;
  for (int c = 0; c < num_class; ++c)
    mx = fmax(mx, score[static_cast<size_t>(c) * n + i]);
  double sum = 0;
  for (int c = 0; c < num_class; ++c) sum += exp(score[static_cast<size_t>(c) * n + i] - mx);
  const int lbl = static_cast<int>(label[i]);
  for (int c = 0; c < num_class; ++c) {
    const size_t k = static_cast<size_t>(c) * n + i;
    const double p = exp(score[k] - mx) / sum;
    g[k] = static_cast<float>(w * (p - (c == lbl ? 1.0 : 0.0)));
    h[k] = static_cast<float>(w * factor * p * (1.0 - p));
  }
}

/*! one-vs-all: per-class sigmoid binary gradients (MulticlassOVA parity). */
__global__ void k_grad_multiclass_ova(const double* __restrict__ score,
                                      const float* __restrict__ label,
                                      const float* __restrict__ weight, int n,
                                      int num_class, double sigmoid,
                                      float* __restrict__ g, float* __restrict__ h) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const double w = weight ? weight[i] : 1.0;
  const int lbl = static_cast<int>(label[i]);
  for (int c = 0; c < num_class; ++c) {
    const size_t k = static_cast<size_t>(c) * n + i;
    const double y = lbl == c ? 1.0 : -1.0;
    const double response = -y * sigmoid / (1.0 + exp(y * sigmoid * score[k]));
    const double ar = fabs(response);
    g[k] = static_cast<float>(response * w);
    h[k] = static_cast<float>(ar * (sigmoid - ar) * w);
  }
}

/*! device pointwise metric reduction: out[0] += Σ w·loss, out[1] += Σ w.
 *  Loss formulas mirror cpp/src/metric.cpp; convert kinds are the objectives'
 *  output transforms. Replaces the per-eval 8B×rows score download
 *  (capability parity: reference src/metric/cuda/cuda_pointwise_metric.cu). */
__global__ void k_metric_pointwise(int kind, double a, int convert_kind,
                                   double convert_param,
                                   const double* __restrict__ score,
                                   const float* __restrict__ label,
                                   const float* __restrict__ weight, int n,
                                   double* __restrict__ out) {
  double loss_sum = 0.0, w_sum = 0.0;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += gridDim.x * blockDim.x) {
    const double w = weight ? weight[i] : 1.0;
    const double y = label[i];
    double p = score[i];
    switch (convert_kind) {
      case 1: p = exp(p); break;
      case 2: p = 1.0 / (1.0 + exp(-convert_param * p)); break;
      case 3: p = 1.0 / (1.0 + exp(-p)); break;
      case 4: p = log1p(exp(p)); break;
      case 5: p = p * fabs(p); break;
      default: break;
    }
    double l = 0.0;
    switch (kind) {
      case 0: l = (y - p) * (y - p); break;                         // l2 / rmse / r2
      case 1: l = fabs(y - p); break;                               // l1
      case 2: { const double dd = y - p; l = dd >= 0 ? a * dd : (a - 1.0) * dd; break; }
      case 3: { const double dd = fabs(y - p);                      // huber
                l = dd <= a ? 0.5 * dd * dd : a * (dd - 0.5 * a); break; }
      case 4: { const double x = fabs(y - p);                       // fair
                l = a * x - a * a * log1p(x / a); break; }
      case 5: { double pp = fmax(p, 1e-10); l = pp - y * log(pp); break; }  // poisson
      case 6: l = fabs((y - p) / fmax(1.0, fabs(y))); break;        // mape
      case 7: { double pp = fmax(p, 1e-10); l = y / pp + log(pp) - 1.0; break; }  // gamma
      case 8: { double pp = fmax(p, 1e-10);                         // gamma_deviance
                l = y <= 1e-10 ? 0.0 : 2.0 * (log(pp / y) + y / pp - 1.0); break; }
      case 9: { double pp = fmax(p, 1e-10);                         // tweedie
                l = -y * pow(pp, 1.0 - a) / (1.0 - a) + pow(pp, 2.0 - a) / (2.0 - a);
                break; }
      case 10: { double pp = fmin(1.0 - 1e-12, fmax(1e-12, p));     // binary_logloss
                 l = y > 0 ? -log(pp) : -log(1.0 - pp); break; }
      case 11: l = ((p > 0.5) != (y > 0)) ? 1.0 : 0.0; break;       // binary_error
      case 12: { double pp = fmin(1.0 - 1e-12, fmax(1e-12, p));     // cross_entropy
                 l = -y * log(pp) - (1.0 - y) * log(1.0 - pp); break; }
      case 13: { const double hhat = log1p(fmax(1e-12, p));         // xentlambda
                 l = y * hhat - p; break; }
    }
    loss_sum += w * l;
    w_sum += w;
  }
  // wave64 reduce, then one atomic per wave
  for (int off = 32; off > 0; off >>= 1) {
    loss_sum += __shfl_down(loss_sum, off);
    w_sum += __shfl_down(w_sum, off);
  }
  if ((threadIdx.x & 63) == 0) {
    atomicAdd(&out[0], loss_sum);
    atomicAdd(&out[1], w_sum);
  }
}

// ------------------------------------------------------ device percentile renew
__device__ __forceinline__ unsigned long long d_ord64(double v) {
  // order-preserving double -> uint64 mapping (sign-magnitude to biased)
  unsigned long long b = static_cast<unsigned long long>(__double_as_longlong(v));
  return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}
__device__ __forceinline__ double d_unord64(unsigned long long u) {
  const unsigned long long b =
      (u & 0x8000000000000000ull) ? (u & 0x7FFFFFFFFFFFFFFFull) : ~u;
  return __longlong_as_double(static_cast<long long>(b));
}

/*! per-leaf α-percentile of residual (label - score) via 64-step bitwise binary
 *  search with block-wide counting — no sort, no extra memory, exact order
 *  statistics. One block per leaf. wmode: 0 unweighted (reference PercentileFun
 *  interpolation), 1 metadata weights, 2 MAPE label weights (w/max(1,|label|)).
 *  Replaces the host download+sort renewal for l1/quantile/mape (capability
 *  parity: reference RenewTreeOutputCUDAKernel_RegressionL1/Quantile,
 *  cuda_regression_objective.cu). */
__global__ void k_renew_percentile(const uint32_t* __restrict__ idx,
                                   const int* __restrict__ leaf_begin,
                                   const int* __restrict__ leaf_cnt, int nl,
                                   const double* __restrict__ score,
                                   const float* __restrict__ label,
                                   const float* __restrict__ weight, int wmode,
                                   double alpha, double* __restrict__ out) {
  const int l = blockIdx.x;
  if (l >= nl) return;
  const int begin = leaf_begin[l];
  const int cnt = leaf_cnt[l];
  if (cnt <= 0) return;  // keep the existing output
  __shared__ double s_red[4];
  __shared__ unsigned long long s_bound[2];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int nwave = blockDim.x >> 6;

  auto residual = [&](int j) -> double {
    const uint32_t row = idx[begin + j];
    return static_cast<double>(label[row]) - score[row];
  };
  auto wgt = [&](int j) -> double {
    const uint32_t row = idx[begin + j];
    double w = weight ? weight[row] : 1.0;
    if (wmode == 2) w /= fmax(1.0, fabs(static_cast<double>(label[row])));
    return w;
  };
  // block-wide sum of a per-thread double
  auto block_sum = [&](double v) -> double {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
    if ((tid & 63) == 0) s_red[wave] = v;
    __syncthreads();
    double t = 0;
    for (int wv = 0; wv < nwave; ++wv) t += s_red[wv];
    __syncthreads();
    return t;
  };

  if (wmode == 0) {
    // k-th order statistic: smallest ordinal m with count(res <= m) >= k+1
    auto kth = [&](int k) -> double {
      unsigned long long lo = 0ull, hi = ~0ull;
      while (lo < hi) {
        const unsigned long long mid = lo + ((hi - lo) >> 1);
        double c = 0;
        for (int j = tid; j < cnt; j += blockDim.x)
          if (d_ord64(residual(j)) <= mid) c += 1.0;
        const double total = block_sum(c);
        if (tid == 0) s_bound[0] = total >= static_cast<double>(k + 1) ? 1ull : 0ull;
        __syncthreads();
        const bool le = s_bound[0] != 0ull;
        __syncthreads();
        if (le) hi = mid;
        else lo = mid + 1ull;
      }
      return d_unord64(lo);
    };
    const double pos = alpha * (cnt - 1);
    const int klo = static_cast<int>(pos);
    const int khi = klo + 1 < cnt ? klo + 1 : cnt - 1;
    const double frac = pos - klo;
    const double vlo = kth(klo);
    const double vhi = khi == klo ? vlo : kth(khi);
    if (tid == 0) out[l] = vlo * (1.0 - frac) + vhi * frac;
  } else {
    double wloc = 0;
    for (int j = tid; j < cnt; j += blockDim.x) wloc += wgt(j);
    const double total_w = block_sum(wloc);
    const double target = alpha * total_w;
    unsigned long long lo = 0ull, hi = ~0ull;
    while (lo < hi) {
      const unsigned long long mid = lo + ((hi - lo) >> 1);
      double c = 0;
      for (int j = tid; j < cnt; j += blockDim.x)
        if (d_ord64(residual(j)) <= mid) c += wgt(j);
      const double cum = block_sum(c);
      if (tid == 0) s_bound[0] = cum >= target ? 1ull : 0ull;
      __syncthreads();
      const bool le = s_bound[0] != 0ull;
      __syncthreads();
      if (le) hi = mid;
      else lo = mid + 1ull;
    }
    if (tid == 0) out[l] = d_unord64(lo);
  }
}

/*! lambdarank NDCG gradients: one block per query. Scores are argsorted in LDS
 *  (bitonic, padded to a power of two <= 1024); pairwise lambdas accumulate into
 *  LDS grad/hess; optional per-query normalization. Capability parity: the
 *  reference's GetGradientsKernel_LambdarankNDCG (fresh wave64 implementation). */
__global__ void __launch_bounds__(256) k_grad_lambdarank(
    const double* __restrict__ score, const float* __restrict__ label,
    const int* __restrict__ qb, int num_queries, const double* __restrict__ inv_max_dcg,
    const double* __restrict__ label_gain, double sigmoid, int trunc, int norm,
    float* __restrict__ out_g, float* __restrict__ out_h) {
  __shared__ float s_score[1024];
  __shared__ short s_idx[1024];
  __shared__ float s_g[1024], s_h[1024];
  __shared__ double s_suml[4];
  const int q = blockIdx.x;
  if (q >= num_queries) return;
  const int start = qb[q];
  const int cnt = qb[q + 1] - start;
  if (cnt > 1024) {
    // the host pre-check refuses such datasets; reaching here means stale g/h
    // would silently corrupt training — trap instead of returning
    __builtin_trap();
  }
  int pow2 = 1;
  while (pow2 < cnt) pow2 <<= 1;
  for (int i = threadIdx.x; i < pow2; i += blockDim.x) {
    s_score[i] = i < cnt ? static_cast<float>(score[start + i]) : -3.0e38f;
    s_idx[i] = static_cast<short>(i);
    if (i < cnt) {
      s_g[i] = 0.0f;
      s_h[i] = 0.0f;
    }
  }
  __syncthreads();
  // bitonic sort descending by score
  for (int ksz = 2; ksz <= pow2; ksz <<= 1) {
    for (int j = ksz >> 1; j > 0; j >>= 1) {
      for (int i = threadIdx.x; i < pow2; i += blockDim.x) {
        const int ixj = i ^ j;
        if (ixj > i) {
          const bool up = (i & ksz) == 0;  // descending overall
          const bool swap = up ? (s_score[i] < s_score[ixj]) : (s_score[i] > s_score[ixj]);
          if (swap) {
            const float ts = s_score[i];
            s_score[i] = s_score[ixj];
            s_score[ixj] = ts;
            const short ti = s_idx[i];
            s_idx[i] = s_idx[ixj];
            s_idx[ixj] = ti;
          }
        }
      }
      __syncthreads();
    }
  }
  const double imd = inv_max_dcg[q];
  if (imd <= 0) {
    for (int i = threadIdx.x; i < cnt; i += blockDim.x) {
      out_g[start + i] = 0.0f;
      out_h[start + i] = 0.0f;
    }
    return;
  }
  const int t = min(trunc, cnt);
  const float best = s_score[0];
  const float worst = s_score[cnt - 1];
  double local_suml = 0.0;
  // pair (i, j) with i < t, i < j < cnt, strided over threads
  const int64_t n_pairs = static_cast<int64_t>(t) * cnt;
  for (int64_t p = threadIdx.x; p < n_pairs; p += blockDim.x) {
    const int i = static_cast<int>(p / cnt);
    const int j = static_cast<int>(p % cnt);
    if (j <= i) continue;
    const int di = s_idx[i], dj = s_idx[j];
    const float li = label[start + di], lj = label[start + dj];
    if (li == lj) continue;
    int hi_rank, lo_rank, hi_doc, lo_doc;
    if (li > lj) {
      hi_rank = i; lo_rank = j; hi_doc = di; lo_doc = dj;
    } else {
      hi_rank = j; lo_rank = i; hi_doc = dj; lo_doc = di;
    }
    const double high_gain = label_gain[static_cast<int>(label[start + hi_doc])];
    const double low_gain = label_gain[static_cast<int>(label[start + lo_doc])];
    const double ds = score[start + hi_doc] - score[start + lo_doc];
    const double high_disc = 1.0 / log2(2.0 + hi_rank);
    const double low_disc = 1.0 / log2(2.0 + lo_rank);
    double delta = fabs((high_gain - low_gain) * (high_disc - low_disc) * imd);
    if (norm && best != worst) delta /= (0.01 + fabs(ds));
    double p_lambda = 1.0 / (1.0 + exp(sigmoid * ds));
    double p_hess = p_lambda * (1.0 - p_lambda);
    p_lambda *= -sigmoid * delta;
    p_hess *= sigmoid * sigmoid * delta;
    atomicAdd(&s_g[hi_doc], static_cast<float>(p_lambda));
    atomicAdd(&s_h[hi_doc], static_cast<float>(p_hess));
    atomicAdd(&s_g[lo_doc], static_cast<float>(-p_lambda));
    atomicAdd(&s_h[lo_doc], static_cast<float>(p_hess));
    local_suml -= 2.0 * p_lambda;
  }
  // block reduce sum_lambdas
  for (int d = 32; d > 0; d >>= 1) local_suml += __shfl_down(local_suml, d);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) s_suml[wave] = local_suml;
  __syncthreads();
  double suml = 0;
  for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) suml += s_suml[w];
  double nf = 1.0;
  if (norm && suml > 0) nf = log2(1.0 + suml) / suml;
  for (int i = threadIdx.x; i < cnt; i += blockDim.x) {
    out_g[start + i] = static_cast<float>(s_g[i] * nf);
    out_h[start + i] = static_cast<float>(s_h[i] * nf);
  }
}

__global__ void k_score_add_const(double* score, int n, double v) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) score[i] += v;
}

/*! score[row] += leaf_output; segments sorted by begin for the binary search. */
__global__ void k_score_update(const uint32_t* __restrict__ idx,
                               const int* __restrict__ sorted_begin, int num_leaves,
                               int used_cnt, const double* __restrict__ leaf_out,
                               double* __restrict__ score) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= used_cnt) return;
  int lo = 0, hi = num_leaves - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (i >= sorted_begin[mid]) lo = mid;
    else hi = mid - 1;
  }
  score[idx[i]] += leaf_out[lo];
}

/*! linear-leaf score update: score[row] += const + coeff . raw[path feats]
 *  (NaN in any path feature falls back to the piecewise-constant output, like
 *  Tree::LeafOutputLinear). Arrays are in SORTED-leaf order (sorted_begin). */
__global__ void k_score_update_linear(
    const uint32_t* __restrict__ idx, const int* __restrict__ sorted_begin,
    int num_leaves, int used_cnt, const double* __restrict__ leaf_const,
    const double* __restrict__ leaf_fallback, const int* __restrict__ coeff_off,
    const int* __restrict__ coeff_cnt, const int* __restrict__ feat_flat,
    const double* __restrict__ coeff_flat, const float* __restrict__ raw,
    int num_data, double* __restrict__ score) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= used_cnt) return;
  int lo = 0, hi = num_leaves - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (i >= sorted_begin[mid]) lo = mid;
    else hi = mid - 1;
  }
  const uint32_t r = idx[i];
  const int k = coeff_cnt[lo];
  double out = leaf_const[lo];
  bool nan = false;
  const int off = coeff_off[lo];
  for (int j = 0; j < k; ++j) {
    const float v = raw[static_cast<size_t>(feat_flat[off + j]) * num_data + r];
    if (isnan(v)) { nan = true; break; }
    out += coeff_flat[off + j] * v;
  }
  score[r] += (nan || k == 0) ? leaf_fallback[lo] : out;
}

/*! per-leaf weighted Gram matrix (upper triangle) + rhs for the linear-tree
 *  post-pass: A += h z z^T, b += -g z over the leaf rows, z = (raw path feats, 1).
 *  Capability parity: reference LinearTreeLearner::CalculateLinear inner loops,
 *  computed on device instead of the host row loop. */
__global__ void k_linear_gram(const uint32_t* __restrict__ idx, int begin, int cnt,
                              const float* __restrict__ raw, int num_data,
                              const int* __restrict__ feats, int k,
                              const float* __restrict__ g, const float* __restrict__ h,
                              double* __restrict__ A, double* __restrict__ b,
                              int* __restrict__ nan_flag) {
  extern __shared__ double s_acc[];  // dim*dim (upper) + dim
  const int dim = k + 1;
  const int n_elem = dim * dim + dim;
  for (int e = threadIdx.x; e < n_elem; e += blockDim.x) s_acc[e] = 0.0;
  __syncthreads();
  double z[33];  // dim capped host-side at 33 (32 features + intercept)
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < cnt;
       i += gridDim.x * blockDim.x) {
    const uint32_t r = idx[begin + i];
    bool nan = false;
    for (int j = 0; j < k; ++j) {
      const float v = raw[static_cast<size_t>(feats[j]) * num_data + r];
      if (isnan(v)) { nan = true; break; }
      z[j] = v;
    }
    if (nan) {
      atomicOr(nan_flag, 1);
      continue;
    }
    z[k] = 1.0;
    const double hv = h[r];
    const double gv = g[r];
    for (int a = 0; a < dim; ++a) {
      const double hza = hv * z[a];
      for (int c2 = a; c2 < dim; ++c2) atomicAdd(&s_acc[a * dim + c2], hza * z[c2]);
      atomicAdd(&s_acc[dim * dim + a], -gv * z[a]);
    }
  }
  __syncthreads();
  for (int e = threadIdx.x; e < dim * dim; e += blockDim.x)
    if (s_acc[e] != 0.0) atomicAdd(&A[e], s_acc[e]);
  for (int e = threadIdx.x; e < dim; e += blockDim.x)
    if (s_acc[dim * dim + e] != 0.0) atomicAdd(&b[e], s_acc[dim * dim + e]);
}

/*! device tree walk over column bins (out-of-bag score update under bagging).
 *  is_cat + cat_mask (4 words/node, bin-space bitset) route categorical splits;
 *  the optional linear arrays (TREE-leaf order) evaluate linear leaves from raw
 *  feature values, falling back to the constant output on NaN path features. */
template <typename BIN_T = uint8_t>
__global__ void k_tree_predict_add(const BIN_T* __restrict__ cols, int num_data,
                                   const int* __restrict__ split_feat,
                                   const int* __restrict__ thr_bin,
                                   const int* __restrict__ left_child,
                                   const int* __restrict__ right_child,
                                   const int* __restrict__ nan_bin,
                                   const uint8_t* __restrict__ default_left,
                                   const uint8_t* __restrict__ is_cat,
                                   const unsigned long long* __restrict__ cat_mask,
                                   const double* __restrict__ leaf_out,
                                   const int* __restrict__ lin_off,
                                   const int* __restrict__ lin_cnt,
                                   const int* __restrict__ lin_feat,
                                   const double* __restrict__ lin_coeff,
                                   const double* __restrict__ lin_const,
                                   const float* __restrict__ raw,
                                   const uint32_t* __restrict__ rows, int n,
                                   double* __restrict__ score) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const uint32_t r = rows[i];
  int node = 0;
  while (node >= 0) {
    const int f = split_feat[node];
    const int b = cols[static_cast<size_t>(f) * num_data + r];
    if (is_cat != nullptr && is_cat[node]) {
      const unsigned long long w = cat_mask[4 * node + (b >> 6)];
      node = ((w >> (b & 63)) & 1ull) ? left_child[node] : right_child[node];
    } else if (nan_bin[node] >= 0 && b == nan_bin[node]) {
      node = default_left[node] ? left_child[node] : right_child[node];
    } else {
      node = b <= thr_bin[node] ? left_child[node] : right_child[node];
    }
  }
  const int leaf = ~node;
  if (lin_cnt != nullptr) {
    const int k = lin_cnt[leaf];
    double out = lin_const[leaf];
    bool nan = false;
    const int off = lin_off[leaf];
    for (int j = 0; j < k; ++j) {
      const float v = raw[static_cast<size_t>(lin_feat[off + j]) * num_data + r];
      if (isnan(v)) { nan = true; break; }
      out += lin_coeff[off + j] * v;
    }
    score[r] += (nan || k == 0) ? leaf_out[leaf] : out;
  } else {
    score[r] += leaf_out[leaf];
  }
}

}  // namespace hipk

// ------------------------------------------------------------------ RAII device buffer
template <typename T>
struct DevBuf {
  T* ptr = nullptr;
  size_t n = 0;
  DevBuf() = default;
  DevBuf(const DevBuf&) = delete;
  DevBuf& operator=(const DevBuf&) = delete;
  void Alloc(size_t count) {
    if (count <= n) return;
    Free();
    if (count > 0) HIP_OK(hipMalloc(&ptr, count * sizeof(T)));
    n = count;
  }
  void Free() {
    if (ptr) hipFree(ptr);
    ptr = nullptr;
    n = 0;
  }
  ~DevBuf() { Free(); }
};

// ------------------------------------------------------------------ the learner
class HIPTreeLearner : public TreeLearner {
 public:
  explicit HIPTreeLearner(const Config* config) : config_(config) {}
  ~HIPTreeLearner() override {
    if (h_winner_) hipHostFree(h_winner_);
    if (stream_) hipStreamDestroy(stream_);
  }

  bool IsHIPLearner() const override { return true; }
  /*! per-learner comm override (single-process num_gpu shard learners); the
   *  process-wide singleton (LGBM_GPUNetworkInit) is the default. */
  void SetCommClique(InProcClique* c, int rank) {
    own_comm_.clique = c;
    own_comm_.rank = rank;
    use_own_comm_ = true;
  }
  void SetCommRccl(ncclComm_t c, int world, int rank) {
    own_comm_.comm = c;
    own_comm_.world = world;
    own_comm_.rank = rank;
    use_own_comm_ = true;
  }
  bool DeviceObjectiveSupported(const std::string& name) const override {
    static const char* kRegression[] = {"regression", "regression_l1", "huber", "fair",
                                        "poisson",    "quantile",      "mape",  "gamma",
                                        "tweedie"};
    for (const char* r : kRegression) {
      // reg_sqrt transforms labels host-side at objective Init; keep host path
      if (name == r) return !config_->reg_sqrt;
    }
    if (name == "binary" || name == "cross_entropy" || name == "cross_entropy_lambda" ||
        name == "multiclass" || name == "multiclassova")
      return true;
    if (name == "lambdarank") return rank_ok_;
    return false;
  }
  bool DeviceEvalPointwise(int loss_kind, double loss_a, int convert_kind,
                           double convert_param, double* out_sum,
                           double* out_wsum) override;
  /*! kernel-level unit-test probe: run the root k_hist on device and compare
   *  bin-for-bin against the fp64 host histogram oracle
   *  (Dataset::ConstructHistograms). Returns the max relative error. */
  double DebugRootHistMaxRelErr(const score_t* g, const score_t* h);

  void Init(const Dataset* train_data, bool is_constant_hessian) override;
  void ResetTrainingData(const Dataset* train_data) override {
    Init(train_data, is_constant_hessian_);
  }
  void ResetConfig(const Config* config) override { config_ = config; }
  Tree* Train(const score_t* gradients, const score_t* hessians, bool is_first) override;
  void SetBaggingData(const Dataset* subset, const data_size_t* used_indices,
                      data_size_t num_data) override {
    if (subset != nullptr) Log::Fatal("HIP learner does not use dataset-subset bagging");
    bag_indices_ = used_indices;
    bag_cnt_ = num_data;
  }
  void AddPredictionToScore(const Tree* tree, double* out_score) override;
  void RenewTreeOutput(Tree* tree, const ObjectiveFunction* obj,
                       std::function<double(const label_t*, int)>, data_size_t,
                       const data_size_t*, data_size_t, const double* train_score) override;

  void DeviceBoosting(const ObjectiveFunction* obj) override;
  void DeviceAddInitScore(double v) override {
    hipLaunchKernelGGL(hipk::k_score_add_const, dim3((num_data_ + 255) / 256), dim3(256), 0,
                       stream_, ScorePtr(), num_data_, v);
  }
  void SetClassOffset(int class_id) override {
    cur_class_ = class_id < num_class_score_ ? class_id : 0;
  }
  void DownloadTrainScore(double* dst) override {
    HIP_OK(hipStreamSynchronize(stream_));
    HIP_OK(hipMemcpy(dst, d_score_.ptr,
                     sizeof(double) * num_data_ * num_class_score_, hipMemcpyDeviceToHost));
  }
  void UploadTrainScore(const double* src) override {
    HIP_OK(hipStreamSynchronize(stream_));
    HIP_OK(hipMemcpy(d_score_.ptr, src,
                     sizeof(double) * num_data_ * num_class_score_, hipMemcpyHostToDevice));
  }

 private:
  void UploadGradients(const score_t* g, const score_t* h);
  void LaunchHist(const int* leafA_ptr, int leafB_from_counters, int blocks, bool zero_spare = true);
  void ReduceSpareHist(int spare_slot);
  void LaunchBestSplit(const int* leafA_ptr, int leafB_from_counters);
  /*! balanced contiguous feature-block ownership for reduce-scatter mode */
  bool SetupOwnership(int world, int rank);
  /*! allgather per-rank winners + deterministic global argmax (ownership mode) */
  void SyncGlobalWinner();
  /*! hipk::GradKind for a pointwise objective name, or -1 */
  int PointwiseGradKind(const std::string& name) const {
    if (name == "regression_l1") return hipk::kGradL1;
    if (name == "huber") return hipk::kGradHuber;
    if (name == "fair") return hipk::kGradFair;
    if (name == "poisson") return hipk::kGradPoisson;
    if (name == "quantile") return hipk::kGradQuantile;
    if (name == "mape") return hipk::kGradMape;
    if (name == "gamma") return hipk::kGradGamma;
    if (name == "tweedie") return hipk::kGradTweedie;
    if (name == "cross_entropy") return hipk::kGradXent;
    if (name == "cross_entropy_lambda") return hipk::kGradXentLambda;
    return -1;
  }
  double PointwiseGradParam(const std::string& name) const {
    if (name == "huber" || name == "quantile") return config_->alpha;
    if (name == "fair") return config_->fair_c;
    if (name == "poisson") return config_->poisson_max_delta_step;
    if (name == "tweedie") return config_->tweedie_variance_power;
    return 0.0;
  }
  int HistBlocksFor(int approx_cnt) const {
    // LDS atomic throughput is per-CU: spread even small leaves over many blocks
    // (~256 rows each); cap so the per-block flush stays amortized at the root.
    int b = (std::max(approx_cnt, 1) + 255) / 256;
    return std::min(2048, std::max(1, b));
  }

  const Config* config_;
  const Dataset* train_data_ = nullptr;
  bool is_constant_hessian_ = false;
  hipStream_t stream_ = nullptr;

  int num_data_ = 0;
  int nf_ = 0;
  int total_bins_ = 0;
  int row_stride_ = 0;
  int n_copies_ = 4;  // LDS histogram privatization factor
  int lds_budget_used_ = 80 * 1024;  // chosen by the partition planner
  std::vector<hipk::FeatMeta> feat_meta_host_;
  std::vector<std::pair<int, int>> feat_partitions_;
  std::vector<std::pair<int, int>> part_bin_range_;

  DevBuf<uint8_t> d_rows_;
  DevBuf<uint8_t> d_cols_;
  DevBuf<hipk::FeatMeta> d_feat_meta_;
  DevBuf<float> d_grad_, d_hess_;
  DevBuf<int32_t> d_grad_packed_;
  DevBuf<float> d_grad_absmax_, d_grad_scales_;
  // lambdarank (device ranking objective)
  bool rank_ok_ = false;
  DevBuf<int> d_qb_;
  DevBuf<double> d_inv_max_dcg_, d_label_gain_;
  int num_queries_ = 0;
  DevBuf<double> d_score_;
  DevBuf<float> d_label_, d_weight_;
  DevBuf<uint32_t> d_idx_, d_idx_tmp_;
  DevBuf<uint8_t> d_marks_;
  DevBuf<int> d_block_cnt_, d_block_loff_, d_block_roff_;
  DevBuf<int> d_ctr_;
  DevBuf<int64_t> d_gbuf_;
  DevBuf<float> d_hist_;
  DevBuf<hipk::SplitRec> d_feat_best_;
  DevBuf<hipk::SplitRec> d_leaf_best_;
  DevBuf<hipk::SplitRec> d_winner_;
  DevBuf<int> d_winner_leaf_;
  DevBuf<int> d_counters_;      // [num_leaves, split_log_index]
  DevBuf<int> d_root_leaf_, d_minus1_;
  DevBuf<hipk::LogEntry> d_split_log_;
  DevBuf<hipk::LeafStat> d_leaf_stats_;
  DevBuf<double> d_leaf_bounds_;   // [2*num_leaves] monotone output bounds
  DevBuf<int8_t> d_mono_;          // per inner feature, only when constraints set
  DevBuf<unsigned long long> d_leaf_branch_;  // [4*num_leaves] branch-feature bitmasks (256-bit)
  DevBuf<unsigned long long> d_group_masks_;  // interaction groups (inner-feature bits)
  int n_interaction_groups_ = 0;
  DevBuf<int8_t> d_feat_mask_;
  DevBuf<float> d_cegb_coupled_, d_cegb_lazy_;  // per inner feature (CEGB)
  bool use_cegb_ = false;
  // forcedsplits: host-precomputed per-iteration schedule (leaf,-1 = no force)
  struct ForcedStep { int leaf = -1; int feature = -1; int bin = -1; };
  std::unique_ptr<ForcedNode> forced_root2_;
  std::vector<ForcedStep> forced_sched_;
  void LaunchForcedWinner(const ForcedStep& fs);
  // linear trees (device Gram post-pass over raw features on the branch path)
  bool linear_ = false;
  DevBuf<float> d_raw_;
  DevBuf<int> d_lin_feats_;
  DevBuf<double> d_lin_A_, d_lin_b_;
  DevBuf<int> d_lin_nan_;
  DevBuf<double> d_lin_const_;
  DevBuf<int> d_lin_coeff_off_, d_lin_coeff_cnt_, d_lin_feat_flat_;
  DevBuf<double> d_lin_coeff_flat_;
  void CalculateLinearDevice(Tree* tree);
  DevBuf<int> d_leaf_begin_, d_leaf_cnt_, d_leaf_slot_;
  DevBuf<int> d_sorted_begin_;
  DevBuf<double> d_leaf_out_;
  DevBuf<uint32_t> d_cat_bits_;
  DevBuf<int> d_tw_feat_, d_tw_thr_, d_tw_left_, d_tw_right_, d_tw_nan_;
  DevBuf<uint8_t> d_tw_dl_;
  DevBuf<double> d_tw_out_;
  DevBuf<uint32_t> d_oob_;
  DevBuf<uint8_t> d_tw_iscat_;             // per-node categorical flags (OOB walk)
  DevBuf<unsigned long long> d_tw_cat_;    // [4*ni] bin-space cat masks (OOB walk)
  DevBuf<int> d_twl_off_, d_twl_cnt_, d_twl_feat_;   // linear arrays, TREE-leaf order
  DevBuf<double> d_twl_coeff_, d_twl_const_;

  hipk::SplitRec* h_winner_ = nullptr;  // pinned staging (legacy; log path reads below)
  int* h_winner_leaf_ = nullptr;
  std::vector<hipk::LogEntry> host_log_;

  bool grads_on_device_ = false;
  bool rows16_ = false;   // uint16 bins (max_bin > 256)
  bool hist_dp_ = false;  // gpu_use_dp: fp64 histogram accumulation end to end
  bool quantized_ = false;
  bool coop_launch_ = false;   // fused cooperative partition kernel available
  bool use_mono_ = false;      // monotone constraints active (bounds tracked on device)
  uint32_t bynode_seed_ = 0x1234ABCDu;  // per-tree component of the device sampling hash
  int num_class_score_ = 1;   // classes in the device score buffer
  int cur_class_ = 0;         // class selected by SetClassOffset
  double* ScorePtr() { return d_score_.ptr + static_cast<size_t>(cur_class_) * num_data_; }
  float* GradPtr() { return d_grad_.ptr + static_cast<size_t>(cur_class_) * num_data_; }
  float* HessPtr() { return d_hess_.ptr + static_cast<size_t>(cur_class_) * num_data_; }
  int quant_levels_ = 2;
  uint32_t quant_seed_ = 0x9E3779B9u;
  bool weights_present_ = false;
  const data_size_t* bag_indices_ = nullptr;
  data_size_t bag_cnt_ = 0;
  data_size_t used_cnt_ = 0;

  std::vector<int> leaf_begin_, leaf_cnt_;   // host mirrors (exact after tree download)
  std::vector<int> approx_cnt_;              // launch-sizing hints during the tree loop
  Random feature_rng_{0};
  std::vector<int8_t> feat_mask_host_;

  // distributed state (multi-GPU data-parallel)
  bool dist_ = false;       // any cross-rank transport active this tree
  bool own_scan_ = false;   // reduce-scatter + per-rank feature-ownership scan
  int own_fb_ = 0, own_fe_ = 0;            // this rank's owned feature range
  int own_world_ = 0;                      // world the ownership plan was built for
  std::vector<size_t> own_off_, own_cnt_;  // per-rank hist float offsets/counts
  DevBuf<hipk::LogEntry> d_wire_my_, d_wire_all_;
  DevBuf<double> d_eval_out_;  // [loss_sum, weight_sum] device metric reduction
  GpuComm own_comm_;           // per-learner transport (num_gpu shard learners)
  bool use_own_comm_ = false;
  GpuComm& Comm() { return use_own_comm_ ? own_comm_ : GpuComm::Get(); }

  static constexpr int kHistBlock = 256;
  /*! hist-kernel workgroup size (k_hist is blockDim-agnostic; partition kernels
   *  stay at kHistBlock). MIGBM_HIST_THREADS overrides for experiments. */
  static int HistThreads() {
    static int v = [] {
      const char* e = getenv("MIGBM_HIST_THREADS");
      int t = e ? atoi(e) : 256;
      return t < 64 ? 64 : (t > 1024 ? 1024 : t);
    }();
    return v;
  }
  // 64 KB is the no-opt-in workgroup LDS limit; CDNA4 physically has 160 KB per
  // CU and >64 KB dynamic LDS is enabled per-kernel via hipFuncSetAttribute.
  // MIGBM_LDS_BUDGET (bytes) overrides for experiments.
  static int LdsBudget() {
    static int v = [] {
      const char* e = getenv("MIGBM_LDS_BUDGET");
      // default 80 KB: guarantees >=2 workgroups per CU (160 KB physical) while
      // letting the headline config run 4 privatized copies in ONE feature
      // partition. Measured on MI355X @10Mx28: 34.8 -> 30.0 ms/iter vs the 64 KB
      // no-opt-in budget (same box; less same-address LDS-atomic serialization).
      return e ? atoi(e) : 80 * 1024;
    }();
    return v;
  }
};

void HIPTreeLearner::Init(const Dataset* train_data, bool is_constant_hessian) {
  train_data_ = train_data;
  is_constant_hessian_ = is_constant_hessian;
  num_data_ = train_data->num_data();
  nf_ = train_data->num_features();
  total_bins_ = train_data->num_total_bin();
  feature_rng_ = Random(config_->feature_fraction_seed);

  if (!stream_) HIP_OK(hipStreamCreate(&stream_));
  if (!h_winner_) {
    HIP_OK(hipHostMalloc(reinterpret_cast<void**>(&h_winner_),
                         sizeof(hipk::SplitRec) + sizeof(int)));
    h_winner_leaf_ = reinterpret_cast<int*>(h_winner_ + 1);
  }

  feat_meta_host_.resize(nf_);
  for (int f = 0; f < nf_; ++f) {
    const BinMapper* m = train_data->FeatureBinMapper(f);
    if (m->bin_type() == BinType::kCategorical && m->num_bin() > 256) {
      Log::Warning("HIP learner evaluates categorical feature %d with one-hot splits "
                   "only (%d categories exceed the 256-bin device sorted-subset scan)",
                   train_data->RealFeatureIndex(f), m->num_bin());
    }
    feat_meta_host_[f] = {static_cast<int>(train_data->hist_offset(f)), m->num_bin(),
                          m->num_numeric_bin(), m->nan_bin(),
                          m->bin_type() == BinType::kCategorical ? 1 : 0};
  }
  d_feat_meta_.Alloc(nf_);
  HIP_OK(hipMemcpy(d_feat_meta_.ptr, feat_meta_host_.data(), sizeof(hipk::FeatMeta) * nf_,
                   hipMemcpyHostToDevice));
  rows16_ = false;  // must be known before LDS partition planning
  for (int f = 0; f < nf_; ++f) rows16_ = rows16_ || feat_meta_host_[f].num_bin > 256;
  hist_dp_ = config_->gpu_use_dp;  // fp64 LDS doubles the partition bin cost

  // LDS feature partitioning with privatized copies; partitions are 16-feature aligned
  // so row bytes load as whole uint4 chunks. Shrink the copy count if bins are too many.
  n_copies_ = 4;
  auto build_partitions = [&](int copies, int budget) -> bool {
    feat_partitions_.clear();
    part_bin_range_.clear();
    const int max_bins =
        budget / ((2 * copies + 2) * (hist_dp_ ? sizeof(double) : sizeof(float)));
    int begin = 0;
    while (begin < nf_) {
      int end = begin;
      int bins = 0;
      while (end < nf_ && end - begin < 256) {
        // advance in 16-feature blocks so uint8 row bytes load as whole uint4
        // chunks (uint16 rows read per element: single-feature granularity)
        int blk_end = std::min(nf_, end + (rows16_ ? 1 : 16));
        int blk_bins = 0;
        for (int f = end; f < blk_end; ++f) blk_bins += feat_meta_host_[f].num_bin;
        if (bins + blk_bins > max_bins && bins > 0) break;
        if (bins + blk_bins > max_bins && bins == 0) return false;  // single block too big
        bins += blk_bins;
        end = blk_end;
      }
      feat_partitions_.emplace_back(begin, end);
      part_bin_range_.emplace_back(feat_meta_host_[begin].bin_off, bins);
      begin = end;
    }
    return true;
  };
  // prefer the FEWEST feature partitions (one pass over the row bytes beats more
  // privatized copies), then the most copies at the smallest LDS budget: wide
  // datasets (e.g. 136 feats) trade occupancy for a single partition — measured
  // 49.3 -> 47.2 ms/iter on MSLR-shaped lambdarank at a 144 KB budget.
  {
    std::vector<int> budgets = {LdsBudget()};
    if (getenv("MIGBM_LDS_BUDGET") == nullptr) {
      for (int b : {112 * 1024, 144 * 1024})
        if (b > LdsBudget()) budgets.push_back(b);
    }
    int best_copies = -1;
    int best_budget = budgets[0];
    size_t best_parts = SIZE_MAX;
    for (int b : budgets) {
      for (int c : {8, 4, 2, 1}) {
        if (c == 8 && config_->use_quantized_grad) continue;  // k_hist_q has no <8>
        if (!build_partitions(c, b)) continue;
        if (feat_partitions_.size() < best_parts) {
          best_parts = feat_partitions_.size();
          best_copies = c;
          best_budget = b;
        }
      }
    }
    if (best_copies < 0) Log::Fatal("Feature bin footprint exceeds LDS budget");
    n_copies_ = best_copies;
    lds_budget_used_ = best_budget;
    build_partitions(n_copies_, best_budget);
    if (best_budget > 64 * 1024) {
      // opt the hist kernels into >64 KB dynamic LDS (CDNA4: up to 160 KB per WG)
      for (const void* k : {reinterpret_cast<const void*>(&hipk::k_hist<8>),
                            reinterpret_cast<const void*>(&hipk::k_hist<4>),
                            reinterpret_cast<const void*>(&hipk::k_hist<2>),
                            reinterpret_cast<const void*>(&hipk::k_hist<1>),
                            reinterpret_cast<const void*>(&hipk::k_hist<8, uint16_t>),
                            reinterpret_cast<const void*>(&hipk::k_hist<4, uint16_t>),
                            reinterpret_cast<const void*>(&hipk::k_hist<2, uint16_t>),
                            reinterpret_cast<const void*>(&hipk::k_hist<1, uint16_t>),
                            reinterpret_cast<const void*>(&hipk::k_hist<8, uint8_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<4, uint8_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<2, uint8_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<1, uint8_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<8, uint16_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<4, uint16_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<2, uint16_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist<1, uint16_t, double>),
                            reinterpret_cast<const void*>(&hipk::k_hist_q<4>),
                            reinterpret_cast<const void*>(&hipk::k_hist_q<2>),
                            reinterpret_cast<const void*>(&hipk::k_hist_q<1>)}) {
        HIP_OK(hipFuncSetAttribute(k, hipFuncAttributeMaxDynamicSharedMemorySize,
                                   lds_budget_used_));
      }
    }
  }

  const auto& view = train_data->GetRowMajorView();
  MIGBM_CHECK(rows16_ == view.is16);
  row_stride_ = view.row_stride;
  if (rows16_) {
    d_rows_.Alloc(view.data16.size() * 2);
    HIP_OK(hipMemcpy(d_rows_.ptr, view.data16.data(), view.data16.size() * 2,
                     hipMemcpyHostToDevice));
  } else {
    d_rows_.Alloc(view.data.size());
    HIP_OK(hipMemcpy(d_rows_.ptr, view.data.data(), view.data.size(),
                     hipMemcpyHostToDevice));
  }
  const size_t col_esize = rows16_ ? 2 : 1;
  d_cols_.Alloc(static_cast<size_t>(nf_) * num_data_ * col_esize);
  {
    // per-feature dense bins on device (EFB bundles are decoded here: the GPU keeps
    // feature-major layouts; 288GB HBM makes the unbundled copy cheap)
    std::vector<uint8_t> colbuf(num_data_ * col_esize);
    for (int f = 0; f < nf_; ++f) {
      const auto& col = train_data->column(train_data->feature_column(f));
      uint8_t* dst = d_cols_.ptr + static_cast<size_t>(f) * num_data_ * col_esize;
      if (rows16_) {
        uint16_t* cb16 = reinterpret_cast<uint16_t*>(colbuf.data());
#pragma omp parallel for schedule(static)
        for (int i = 0; i < num_data_; ++i)
          cb16[i] = static_cast<uint16_t>(train_data->GetBin(i, f));
        HIP_OK(hipMemcpy(dst, colbuf.data(), num_data_ * 2, hipMemcpyHostToDevice));
      } else if (train_data->feature_bundled(f) || col.is_sparse() || col.is4()) {
        // bundled features decode; sparse columns densify (the device layout is
        // dense row-major + col-major — sparsity is a host-memory concern)
#pragma omp parallel for schedule(static)
        for (int i = 0; i < num_data_; ++i)
          colbuf[i] = static_cast<uint8_t>(train_data->GetBin(i, f));
        HIP_OK(hipMemcpy(dst, colbuf.data(), num_data_, hipMemcpyHostToDevice));
      } else {
        HIP_OK(hipMemcpy(dst, col.data8(), num_data_, hipMemcpyHostToDevice));
      }
    }
  }

  // device multiclass objectives fill all classes at once (class-major, stride
  // num_data_); single-class objectives use only the first slice
  num_class_score_ = std::max(1, config_->num_class);
  d_grad_.Alloc(static_cast<size_t>(num_data_) * num_class_score_);
  d_hess_.Alloc(static_cast<size_t>(num_data_) * num_class_score_);
  {
    int dev = 0, coop = 0;
    HIP_OK(hipGetDevice(&dev));
    HIP_OK(hipDeviceGetAttribute(&coop, hipDeviceAttributeCooperativeLaunch, dev));
    // Measured on MI355X/ROCm 7.2: the cooperative fused partition is ~1.4x
    // SLOWER than 4 stream launches (grid-sync dispatch overhead dominates at
    // this kernel size) — opt-in only, kept as a documented negative result.
    coop_launch_ = coop != 0 && getenv("MIGBM_COOP_PARTITION") != nullptr && !rows16_;
  }
  forced_root2_ = ParseForcedSplits(config_->forcedsplits_filename);
  linear_ = config_->linear_tree && train_data->has_raw();
  if (config_->linear_tree && !train_data->has_raw())
    Log::Warning("linear_tree requires raw values; dataset was built without them");
  if (linear_) {
    // raw feature values, feature-major (1 float per cell; 288GB HBM)
    d_raw_.Alloc(static_cast<size_t>(nf_) * num_data_);
    for (int f = 0; f < nf_; ++f) {
      HIP_OK(hipMemcpy(d_raw_.ptr + static_cast<size_t>(f) * num_data_,
                       train_data->raw_column(f), sizeof(float) * num_data_,
                       hipMemcpyHostToDevice));
    }
  }
  use_cegb_ = config_->cegb_tradeoff > 0.0 &&
              (config_->cegb_penalty_split > 0.0 ||
               !config_->cegb_penalty_feature_coupled.empty() ||
               !config_->cegb_penalty_feature_lazy.empty());
  if (use_cegb_) {
    std::vector<float> coupled(nf_, 0.0f), lazy(nf_, 0.0f);
    for (int f = 0; f < nf_; ++f) {
      const int orig = train_data->RealFeatureIndex(f);
      if (orig < static_cast<int>(config_->cegb_penalty_feature_coupled.size()))
        coupled[f] = static_cast<float>(config_->cegb_penalty_feature_coupled[orig]);
      if (orig < static_cast<int>(config_->cegb_penalty_feature_lazy.size()))
        lazy[f] = static_cast<float>(config_->cegb_penalty_feature_lazy[orig]);
    }
    d_cegb_coupled_.Alloc(nf_);
    d_cegb_lazy_.Alloc(nf_);
    HIP_OK(hipMemcpy(d_cegb_coupled_.ptr, coupled.data(), sizeof(float) * nf_,
                     hipMemcpyHostToDevice));
    HIP_OK(hipMemcpy(d_cegb_lazy_.ptr, lazy.data(), sizeof(float) * nf_,
                     hipMemcpyHostToDevice));
  }
  quantized_ = config_->use_quantized_grad;
  if (quantized_ && hist_dp_) {
    Log::Warning("gpu_use_dp with use_quantized_grad runs unquantized (the packed-int "
                 "histogram path is fp32-flush only)");
    quantized_ = false;
  }
  if (quantized_ && rows16_) {
    Log::Warning("use_quantized_grad with max_bin>256 runs unquantized on GPU");
    quantized_ = false;
  }
  quant_levels_ = std::max(1, config_->num_grad_quant_bins / 2);
  if (quantized_) {
    d_grad_packed_.Alloc(num_data_);
    d_grad_absmax_.Alloc(2);
    d_grad_scales_.Alloc(2);
  }
  d_score_.Alloc(static_cast<size_t>(num_data_) * num_class_score_);
  HIP_OK(hipMemset(d_score_.ptr, 0,
                   sizeof(double) * num_data_ * num_class_score_));
  d_label_.Alloc(num_data_);
  HIP_OK(hipMemcpy(d_label_.ptr, train_data->metadata().label(), sizeof(float) * num_data_,
                   hipMemcpyHostToDevice));
  weights_present_ = train_data->metadata().weights() != nullptr;
  if (weights_present_) {
    d_weight_.Alloc(num_data_);
    HIP_OK(hipMemcpy(d_weight_.ptr, train_data->metadata().weights(),
                     sizeof(float) * num_data_, hipMemcpyHostToDevice));
  }
  d_idx_.Alloc(num_data_);
  d_idx_tmp_.Alloc(num_data_);
  d_marks_.Alloc(num_data_);
  d_block_cnt_.Alloc(4097);
  d_block_loff_.Alloc(4097);
  d_block_roff_.Alloc(4097);
  d_ctr_.Alloc(2);
  d_gbuf_.Alloc(2);
  const int nl = config_->num_leaves;
  d_hist_.Alloc(static_cast<size_t>(nl) * total_bins_ * 2 * (hist_dp_ ? 2 : 1));
  d_feat_best_.Alloc(static_cast<size_t>(2) * nf_);
  d_leaf_best_.Alloc(nl);
  d_winner_.Alloc(2);  // rec + trailing winner-leaf int
  d_winner_leaf_.Alloc(1);
  d_counters_.Alloc(2);
  d_root_leaf_.Alloc(1);
  d_minus1_.Alloc(1);
  d_split_log_.Alloc(nl);
  d_leaf_stats_.Alloc(nl);
  d_leaf_bounds_.Alloc(2 * static_cast<size_t>(nl));
  d_leaf_branch_.Alloc(4 * static_cast<size_t>(nl));
  n_interaction_groups_ = 0;
  if (!config_->interaction_constraints.empty() && nf_ > 256)
    Log::Fatal("device interaction constraints support up to 256 features (%d present); "
               "use device_type=cpu for this dataset", nf_);
  if (!config_->interaction_constraints.empty()) {
    // parse "[0,1],[2,3]" original-feature groups into 4-word (256-bit)
    // inner-feature bitmasks (same layout as the categorical cat_mask words)
    std::vector<unsigned long long> masks;
    std::vector<int> orig_to_inner(train_data->num_total_features(), -1);
    for (int f = 0; f < nf_; ++f) orig_to_inner[train_data->RealFeatureIndex(f)] = f;
    std::string sgrp = config_->interaction_constraints;
    size_t pos = 0;
    while ((pos = sgrp.find('[', pos)) != std::string::npos) {
      size_t end = sgrp.find(']', pos);
      if (end == std::string::npos) break;
      unsigned long long msk[4] = {0, 0, 0, 0};
      for (auto& tok : Common::Split(sgrp.substr(pos + 1, end - pos - 1).c_str(), ',')) {
        auto t = Common::Trim(tok);
        if (t.empty()) continue;
        const int orig = atoi(t.c_str());
        if (orig >= 0 && orig < static_cast<int>(orig_to_inner.size()) &&
            orig_to_inner[orig] >= 0) {
          const int f = orig_to_inner[orig];
          msk[f >> 6] |= 1ull << (f & 63);
        }
      }
      if (msk[0] | msk[1] | msk[2] | msk[3])
        for (int w = 0; w < 4; ++w) masks.push_back(msk[w]);
      pos = end + 1;
    }
    if (!masks.empty()) {
      n_interaction_groups_ = static_cast<int>(masks.size() / 4);
      d_group_masks_.Alloc(masks.size());
      HIP_OK(hipMemcpy(d_group_masks_.ptr, masks.data(),
                       sizeof(unsigned long long) * masks.size(), hipMemcpyHostToDevice));
    }
  }
  use_mono_ = false;
  {
    const auto& mcs = config_->monotone_constraints;
    if (!mcs.empty()) {
      std::vector<int8_t> mono_host(nf_, 0);
      for (int f = 0; f < nf_; ++f) {
        const int orig = train_data->RealFeatureIndex(f);
        if (orig < static_cast<int>(mcs.size()))
          mono_host[f] = static_cast<int8_t>(mcs[orig]);
        use_mono_ = use_mono_ || mono_host[f] != 0;
      }
      if (use_mono_) {
        d_mono_.Alloc(nf_);
        HIP_OK(hipMemcpy(d_mono_.ptr, mono_host.data(), nf_, hipMemcpyHostToDevice));
        if (config_->monotone_constraints_method != "basic")
          Log::Warning("device learner enforces monotone constraints with the "
                       "basic policy; use device_type=cpu for "
                       "monotone_constraints_method=%s",
                       config_->monotone_constraints_method.c_str());
      }
    }
  }
  d_feat_mask_.Alloc(nf_);
  d_leaf_begin_.Alloc(nl);
  d_leaf_cnt_.Alloc(nl);
  d_leaf_slot_.Alloc(nl);
  d_sorted_begin_.Alloc(nl);
  d_leaf_out_.Alloc(nl);
  d_cat_bits_.Alloc(16);
  d_tw_feat_.Alloc(nl);
  d_tw_thr_.Alloc(nl);
  d_tw_left_.Alloc(nl);
  d_tw_right_.Alloc(nl);
  d_tw_nan_.Alloc(nl);
  d_tw_dl_.Alloc(nl);
  d_tw_out_.Alloc(nl);
  d_oob_.Alloc(num_data_);

  leaf_begin_.resize(nl);
  leaf_cnt_.resize(nl);
  approx_cnt_.resize(nl);

  // ranking metadata (device lambdarank gradients)
  rank_ok_ = false;
  const data_size_t* qb = train_data->metadata().query_boundaries();
  if (qb != nullptr) {
    num_queries_ = train_data->metadata().num_queries();
    int max_docs = 0;
    for (int q = 0; q < num_queries_; ++q)
      max_docs = std::max<int>(max_docs, qb[q + 1] - qb[q]);
    if (max_docs <= 1024) {
      rank_ok_ = true;
      d_qb_.Alloc(num_queries_ + 1);
      HIP_OK(hipMemcpy(d_qb_.ptr, qb, sizeof(int) * (num_queries_ + 1),
                       hipMemcpyHostToDevice));
      // label-gain table + per-query inverse max DCG (labels are static)
      std::vector<double> lg = config_->label_gain;
      if (lg.empty())
        for (int i = 0; i < 31; ++i) lg.push_back((1u << i) - 1.0);
      d_label_gain_.Alloc(lg.size());
      HIP_OK(hipMemcpy(d_label_gain_.ptr, lg.data(), sizeof(double) * lg.size(),
                       hipMemcpyHostToDevice));
      const label_t* lab = train_data->metadata().label();
      std::vector<double> imd(num_queries_);
      const int trunc = config_->lambdarank_truncation_level;
#pragma omp parallel for schedule(static)
      for (int q = 0; q < num_queries_; ++q) {
        std::vector<double> gains;
        for (data_size_t i = qb[q]; i < qb[q + 1]; ++i)
          gains.push_back(lg[static_cast<int>(lab[i])]);
        std::sort(gains.begin(), gains.end(), std::greater<double>());
        double dcg = 0;
        const int k = std::min<int>(trunc, static_cast<int>(gains.size()));
        for (int i = 0; i < k; ++i) dcg += gains[i] / std::log2(2.0 + i);
        imd[q] = dcg > 0 ? 1.0 / dcg : 0.0;
      }
      d_inv_max_dcg_.Alloc(num_queries_);
      HIP_OK(hipMemcpy(d_inv_max_dcg_.ptr, imd.data(), sizeof(double) * num_queries_,
                       hipMemcpyHostToDevice));
    } else {
      Log::Warning("lambdarank on GPU supports queries up to 1024 docs; the largest "
                   "query has %d -> gradients fall back to the host", max_docs);
    }
  }
  Log::Info("HIP tree learner: %d rows, %d features, %d bins, %zu LDS partitions, "
            "%d hist copies",
            num_data_, nf_, total_bins_, feat_partitions_.size(), n_copies_);
}

void HIPTreeLearner::UploadGradients(const score_t* g, const score_t* h) {
  HIP_OK(hipMemcpyAsync(GradPtr(), g, sizeof(float) * num_data_, hipMemcpyHostToDevice,
                        stream_));
  HIP_OK(hipMemcpyAsync(HessPtr(), h, sizeof(float) * num_data_, hipMemcpyHostToDevice,
                        stream_));
}

void HIPTreeLearner::DeviceBoosting(const ObjectiveFunction* obj) {
  const std::string name = obj->GetName();
  const int n = num_data_;
  const dim3 g((n + 255) / 256), b(256);
  if (name == "binary") {
    double w_pos = config_->scale_pos_weight, w_neg = 1.0;
    if (config_->is_unbalance) {
      data_size_t pos = obj->NumPositiveData();
      data_size_t neg = num_data_ - pos;
      if (pos > 0 && neg > 0) {
        w_pos = pos > neg ? 1.0 : static_cast<double>(neg) / pos;
        w_neg = pos > neg ? static_cast<double>(pos) / neg : 1.0;
      }
    }
    hipLaunchKernelGGL(hipk::k_grad_binary, g, b, 0, stream_, d_score_.ptr, d_label_.ptr,
                       weights_present_ ? d_weight_.ptr : nullptr, n, config_->sigmoid,
                       w_pos, w_neg, d_grad_.ptr, d_hess_.ptr);
  } else if (name == "regression") {
    hipLaunchKernelGGL(hipk::k_grad_l2, g, b, 0, stream_, d_score_.ptr, d_label_.ptr,
                       weights_present_ ? d_weight_.ptr : nullptr, n, d_grad_.ptr,
                       d_hess_.ptr);
  } else if (name == "multiclass") {
    hipLaunchKernelGGL(hipk::k_grad_multiclass, g, b, 0, stream_, d_score_.ptr,
                       d_label_.ptr, weights_present_ ? d_weight_.ptr : nullptr, n,
                       num_class_score_, d_grad_.ptr, d_hess_.ptr);
  } else if (name == "multiclassova") {
    hipLaunchKernelGGL(hipk::k_grad_multiclass_ova, g, b, 0, stream_, d_score_.ptr,
                       d_label_.ptr, weights_present_ ? d_weight_.ptr : nullptr, n,
                       num_class_score_, config_->sigmoid, d_grad_.ptr, d_hess_.ptr);
  } else if (int kind = PointwiseGradKind(name); kind >= 0) {
    hipLaunchKernelGGL(hipk::k_grad_pointwise, g, b, 0, stream_, kind, d_score_.ptr,
                       d_label_.ptr, weights_present_ ? d_weight_.ptr : nullptr, n,
                       PointwiseGradParam(name), d_grad_.ptr, d_hess_.ptr);
  } else if (name == "lambdarank") {
    hipLaunchKernelGGL(hipk::k_grad_lambdarank, dim3(num_queries_), dim3(256), 0, stream_,
                       d_score_.ptr, d_label_.ptr, d_qb_.ptr, num_queries_,
                       d_inv_max_dcg_.ptr, d_label_gain_.ptr, config_->sigmoid,
                       config_->lambdarank_truncation_level,
                       config_->lambdarank_norm ? 1 : 0, d_grad_.ptr, d_hess_.ptr);
  } else {
    Log::Fatal("DeviceBoosting called for unsupported objective %s", name.c_str());
  }
  grads_on_device_ = true;
}

void HIPTreeLearner::LaunchHist(const int* leafA_ptr, int leafB_from_counters,
                                int blocks, bool zero_spare) {
  const int n_elem = total_bins_ * 2;
  const size_t slot_stride = static_cast<size_t>(total_bins_) * 2;
  // zero the spare slot (literal slot 0 for the root). In the split loop the spare
  // was already zeroed by the fused k_finalize, so the launch is skipped.
  if (zero_spare) {
    if (hist_dp_)
      hipLaunchKernelGGL(hipk::k_hist_zero, dim3(16), dim3(256), 0, stream_,
                         reinterpret_cast<double*>(d_hist_.ptr), slot_stride,
                         d_counters_.ptr, leafB_from_counters, 0, leafA_ptr, n_elem);
    else
      hipLaunchKernelGGL(hipk::k_hist_zero, dim3(16), dim3(256), 0, stream_, d_hist_.ptr,
                         slot_stride, d_counters_.ptr, leafB_from_counters, 0, leafA_ptr,
                         n_elem);
  }
  for (size_t pr = 0; pr < feat_partitions_.size(); ++pr) {
    const auto [fb, fe] = feat_partitions_[pr];
    const auto [bin_base, bins] = part_bin_range_[pr];
    if (quantized_) {
      const size_t ldsq = static_cast<size_t>(bins) * (n_copies_ + 1) * sizeof(int);
      switch (n_copies_) {
        case 4:
          hipLaunchKernelGGL(hipk::k_hist_q<4>, dim3(blocks), dim3(kHistBlock), ldsq,
                             stream_, d_rows_.ptr, row_stride_, d_idx_.ptr,
                             d_leaf_begin_.ptr, d_leaf_cnt_.ptr, d_leaf_stats_.ptr,
                             d_leaf_slot_.ptr, leafA_ptr, d_counters_.ptr,
                             leafB_from_counters, d_grad_packed_.ptr, d_grad_scales_.ptr,
                             d_feat_meta_.ptr, fb, fe, bin_base, bins, d_hist_.ptr,
                             slot_stride);
          break;
        case 2:
          hipLaunchKernelGGL(hipk::k_hist_q<2>, dim3(blocks), dim3(kHistBlock), ldsq,
                             stream_, d_rows_.ptr, row_stride_, d_idx_.ptr,
                             d_leaf_begin_.ptr, d_leaf_cnt_.ptr, d_leaf_stats_.ptr,
                             d_leaf_slot_.ptr, leafA_ptr, d_counters_.ptr,
                             leafB_from_counters, d_grad_packed_.ptr, d_grad_scales_.ptr,
                             d_feat_meta_.ptr, fb, fe, bin_base, bins, d_hist_.ptr,
                             slot_stride);
          break;
        default:
          hipLaunchKernelGGL(hipk::k_hist_q<1>, dim3(blocks), dim3(kHistBlock), ldsq,
                             stream_, d_rows_.ptr, row_stride_, d_idx_.ptr,
                             d_leaf_begin_.ptr, d_leaf_cnt_.ptr, d_leaf_stats_.ptr,
                             d_leaf_slot_.ptr, leafA_ptr, d_counters_.ptr,
                             leafB_from_counters, d_grad_packed_.ptr, d_grad_scales_.ptr,
                             d_feat_meta_.ptr, fb, fe, bin_base, bins, d_hist_.ptr,
                             slot_stride);
      }
      continue;
    }
    const size_t lds = static_cast<size_t>(bins) * (2 * n_copies_ + 2) *
                       (hist_dp_ ? sizeof(double) : sizeof(float));
    auto launch_hist = [&](auto bin_tag, auto hist_tag) {
      using BIN_T = decltype(bin_tag);
      using HIST_T = decltype(hist_tag);
      const BIN_T* rp = reinterpret_cast<const BIN_T*>(d_rows_.ptr);
      HIST_T* hb = reinterpret_cast<HIST_T*>(d_hist_.ptr);
      switch (n_copies_) {
        case 8:
          hipLaunchKernelGGL((hipk::k_hist<8, BIN_T, HIST_T>), dim3(blocks), dim3(HistThreads()),
                             lds, stream_, rp, row_stride_, d_idx_.ptr, d_leaf_begin_.ptr,
                             d_leaf_cnt_.ptr, d_leaf_stats_.ptr, d_leaf_slot_.ptr,
                             leafA_ptr, d_counters_.ptr, leafB_from_counters, GradPtr(),
                             HessPtr(), d_feat_meta_.ptr, fb, fe, bin_base, bins,
                             hb, slot_stride);
          break;
        case 4:
          hipLaunchKernelGGL((hipk::k_hist<4, BIN_T, HIST_T>), dim3(blocks), dim3(HistThreads()),
                             lds, stream_, rp, row_stride_, d_idx_.ptr, d_leaf_begin_.ptr,
                             d_leaf_cnt_.ptr, d_leaf_stats_.ptr, d_leaf_slot_.ptr,
                             leafA_ptr, d_counters_.ptr, leafB_from_counters, GradPtr(),
                             HessPtr(), d_feat_meta_.ptr, fb, fe, bin_base, bins,
                             hb, slot_stride);
          break;
        case 2:
          hipLaunchKernelGGL((hipk::k_hist<2, BIN_T, HIST_T>), dim3(blocks), dim3(HistThreads()),
                             lds, stream_, rp, row_stride_, d_idx_.ptr, d_leaf_begin_.ptr,
                             d_leaf_cnt_.ptr, d_leaf_stats_.ptr, d_leaf_slot_.ptr,
                             leafA_ptr, d_counters_.ptr, leafB_from_counters, GradPtr(),
                             HessPtr(), d_feat_meta_.ptr, fb, fe, bin_base, bins,
                             hb, slot_stride);
          break;
        default:
          hipLaunchKernelGGL((hipk::k_hist<1, BIN_T, HIST_T>), dim3(blocks), dim3(HistThreads()),
                             lds, stream_, rp, row_stride_, d_idx_.ptr, d_leaf_begin_.ptr,
                             d_leaf_cnt_.ptr, d_leaf_stats_.ptr, d_leaf_slot_.ptr,
                             leafA_ptr, d_counters_.ptr, leafB_from_counters, GradPtr(),
                             HessPtr(), d_feat_meta_.ptr, fb, fe, bin_base, bins,
                             hb, slot_stride);
      }
    };
    if (rows16_ && hist_dp_) launch_hist(uint16_t{}, double{});
    else if (rows16_) launch_hist(uint16_t{}, float{});
    else if (hist_dp_) launch_hist(uint8_t{}, double{});
    else launch_hist(uint8_t{}, float{});
  }
}

void HIPTreeLearner::ReduceSpareHist(int spare_slot) {
  auto& comm = Comm();
  if (!dist_) return;
  const size_t n = static_cast<size_t>(total_bins_) * 2;
  if (hist_dp_) {
    double* spare = reinterpret_cast<double*>(d_hist_.ptr) +
                    static_cast<size_t>(spare_slot) * n;
    if (own_scan_) comm.ReduceBlocks(spare, own_off_, own_cnt_, stream_);
    else comm.AllReduce(spare, n, stream_);
    return;
  }
  float* spare = d_hist_.ptr + static_cast<size_t>(spare_slot) * n;
  if (own_scan_) {
    // reduce-scatter analogue: rank r receives the global sum of its owned
    // feature block only (matches the reference data-parallel learner's
    // ReduceScatter + owned-feature gain scan, data_parallel_tree_learner.cpp:283-450)
    comm.ReduceBlocks(spare, own_off_, own_cnt_, stream_);
  } else {
    comm.AllReduce(spare, n, stream_);
  }
}

bool HIPTreeLearner::SetupOwnership(int world, int rank) {
  if (world <= 1 || nf_ < world) return false;
  if (own_world_ == world && !own_off_.empty()) return true;  // plan is per-dataset
  // contiguous feature blocks balanced by bin count (hist offsets are contiguous
  // ascending, Dataset::FinishBinMappers)
  std::vector<int> fb(world + 1, nf_);
  fb[0] = 0;
  int f = 0;
  for (int r = 1; r < world; ++r) {
    const double target = static_cast<double>(total_bins_) * r / world;
    while (f < nf_ - (world - r) &&
           feat_meta_host_[f].bin_off + feat_meta_host_[f].num_bin / 2 < target)
      ++f;
    f = std::max(f, fb[r - 1] + 1);  // at least one feature per rank
    fb[r] = f;
  }
  own_off_.assign(world, 0);
  own_cnt_.assign(world, 0);
  for (int r = 0; r < world; ++r) {
    const int b0 = feat_meta_host_[fb[r]].bin_off;
    const int b1 = fb[r + 1] < nf_ ? feat_meta_host_[fb[r + 1]].bin_off : total_bins_;
    own_off_[r] = static_cast<size_t>(b0) * 2;
    own_cnt_[r] = static_cast<size_t>(b1 - b0) * 2;
  }
  own_fb_ = fb[rank];
  own_fe_ = fb[rank + 1];
  own_world_ = world;
  if (!d_wire_my_.ptr) {
    d_wire_my_.Alloc(1);
    d_wire_all_.Alloc(world);
  }
  return true;
}

void HIPTreeLearner::LaunchForcedWinner(const ForcedStep& fs) {
  const size_t slot_stride = static_cast<size_t>(total_bins_) * 2;
  hipk::GainParams p = {};
  p.l1 = config_->lambda_l1;
  p.l2 = config_->lambda_l2;
  p.mds = config_->max_delta_step;
  if (hist_dp_) {
    hipLaunchKernelGGL(hipk::k_force_winner, dim3(1), dim3(64), 0, stream_,
                       reinterpret_cast<const double*>(d_hist_.ptr), slot_stride,
                       d_leaf_slot_.ptr, d_feat_meta_.ptr, d_leaf_stats_.ptr, fs.leaf,
                       fs.feature, fs.bin, p, use_mono_ ? d_mono_.ptr : nullptr,
                       d_leaf_bounds_.ptr, d_winner_.ptr, d_winner_leaf_.ptr);
  } else {
    hipLaunchKernelGGL(hipk::k_force_winner, dim3(1), dim3(64), 0, stream_,
                       const_cast<const float*>(d_hist_.ptr), slot_stride,
                       d_leaf_slot_.ptr, d_feat_meta_.ptr, d_leaf_stats_.ptr, fs.leaf,
                       fs.feature, fs.bin, p, use_mono_ ? d_mono_.ptr : nullptr,
                       d_leaf_bounds_.ptr, d_winner_.ptr, d_winner_leaf_.ptr);
  }
}

void HIPTreeLearner::SyncGlobalWinner() {
  auto& comm = Comm();
  hipLaunchKernelGGL(hipk::k_pack_winner, dim3(1), dim3(1), 0, stream_, d_winner_.ptr,
                     d_winner_leaf_.ptr, d_wire_my_.ptr);
  comm.AllGather(d_wire_my_.ptr, d_wire_all_.ptr, sizeof(hipk::LogEntry), stream_);
  hipLaunchKernelGGL(hipk::k_pick_global_winner, dim3(1), dim3(1), 0, stream_,
                     d_wire_all_.ptr, comm.World(), d_winner_.ptr, d_winner_leaf_.ptr);
}

void HIPTreeLearner::LaunchBestSplit(const int* leafA_ptr, int leafB_from_counters) {
  hipk::GainParams p;
  p.l1 = config_->lambda_l1;
  p.l2 = config_->lambda_l2;
  p.mds = config_->max_delta_step;
  p.min_hess = config_->min_sum_hessian_in_leaf;
  p.min_gain_to_split = config_->min_gain_to_split;
  p.min_data = config_->min_data_in_leaf;
  p.smooth = config_->path_smooth;
  p.bynode_frac = static_cast<float>(config_->feature_fraction_bynode);
  p.cat_l2 = config_->cat_l2;
  p.cat_smooth = config_->cat_smooth;
  p.max_cat_to_onehot = config_->max_cat_to_onehot;
  p.max_cat_threshold = config_->max_cat_threshold;
  p.n_interaction_groups = n_interaction_groups_;
  p.extra_trees = config_->extra_trees ? 1 : 0;
  p.rng_seed = bynode_seed_;
  p.own_fb = own_scan_ ? own_fb_ : 0;
  p.own_fe = own_scan_ ? own_fe_ : nf_;
  p.max_depth = config_->max_depth;
  p.mono_penalty = use_mono_ ? config_->monotone_penalty : 0.0;
  p.cegb_tradeoff = use_cegb_ ? config_->cegb_tradeoff : 0.0;
  p.cegb_split_pen = config_->cegb_penalty_split;
  const size_t slot_stride = static_cast<size_t>(total_bins_) * 2;
  const int ny = leafB_from_counters ? 2 : 1;
  if (hist_dp_) {
    hipLaunchKernelGGL(hipk::k_best_feat, dim3(nf_, ny), dim3(64), 0, stream_,
                       reinterpret_cast<const double*>(d_hist_.ptr), slot_stride,
                       d_leaf_slot_.ptr, d_feat_meta_.ptr, nf_, d_leaf_stats_.ptr,
                       leafA_ptr, d_counters_.ptr, leafB_from_counters, p,
                       feat_mask_host_.empty() ? nullptr : d_feat_mask_.ptr,
                       use_mono_ ? d_mono_.ptr : nullptr, d_leaf_bounds_.ptr,
                       n_interaction_groups_ > 0 ? d_group_masks_.ptr : nullptr,
                       d_leaf_branch_.ptr, use_cegb_ ? d_cegb_coupled_.ptr : nullptr,
                       use_cegb_ ? d_cegb_lazy_.ptr : nullptr, d_feat_best_.ptr);
  } else {
    hipLaunchKernelGGL(hipk::k_best_feat, dim3(nf_, ny), dim3(64), 0, stream_,
                       const_cast<const float*>(d_hist_.ptr), slot_stride,
                       d_leaf_slot_.ptr, d_feat_meta_.ptr, nf_, d_leaf_stats_.ptr,
                       leafA_ptr, d_counters_.ptr, leafB_from_counters, p,
                       feat_mask_host_.empty() ? nullptr : d_feat_mask_.ptr,
                       use_mono_ ? d_mono_.ptr : nullptr, d_leaf_bounds_.ptr,
                       n_interaction_groups_ > 0 ? d_group_masks_.ptr : nullptr,
                       d_leaf_branch_.ptr, use_cegb_ ? d_cegb_coupled_.ptr : nullptr,
                       use_cegb_ ? d_cegb_lazy_.ptr : nullptr, d_feat_best_.ptr);
  }
  (void)ny;
  hipLaunchKernelGGL(hipk::k_best_leaf_overall, dim3(1), dim3(256), 0, stream_,
                     d_feat_best_.ptr, nf_, d_leaf_best_.ptr, leafA_ptr, d_counters_.ptr,
                     leafB_from_counters, d_winner_.ptr, d_winner_leaf_.ptr);
}

Tree* HIPTreeLearner::Train(const score_t* gradients, const score_t* hessians, bool) {
  const int nl = config_->num_leaves;
  bynode_seed_ = bynode_seed_ * 1664525u + 1013904223u +
                 static_cast<uint32_t>(config_->extra_seed +
                                       config_->feature_fraction_seed * 2654435761u);
  auto tree = std::make_unique<Tree>(nl);

  if (!grads_on_device_) UploadGradients(gradients, hessians);
  if (cur_class_ >= num_class_score_ - 1) grads_on_device_ = false;

  feat_mask_host_.clear();
  if (config_->feature_fraction < 1.0) {
    feat_mask_host_.assign(nf_, 0);
    int k = std::max(1, static_cast<int>(nf_ * config_->feature_fraction));
    for (int f : feature_rng_.Sample(nf_, k)) feat_mask_host_[f] = 1;
    HIP_OK(hipMemcpyAsync(d_feat_mask_.ptr, feat_mask_host_.data(), nf_,
                          hipMemcpyHostToDevice, stream_));
  }

  if (bag_indices_ != nullptr && bag_cnt_ > 0) {
    used_cnt_ = bag_cnt_;
    HIP_OK(hipMemcpyAsync(d_idx_.ptr, bag_indices_, sizeof(uint32_t) * bag_cnt_,
                          hipMemcpyHostToDevice, stream_));
  } else {
    used_cnt_ = num_data_;
    hipLaunchKernelGGL(hipk::k_iota, dim3((num_data_ + 255) / 256), dim3(256), 0, stream_,
                       d_idx_.ptr, num_data_);
  }
  if (quantized_) {
    HIP_OK(hipMemsetAsync(d_grad_absmax_.ptr, 0, 2 * sizeof(float), stream_));
    hipLaunchKernelGGL(hipk::k_grad_absmax, dim3(512), dim3(256), 0, stream_, d_idx_.ptr,
                       static_cast<int>(used_cnt_), GradPtr(), HessPtr(),
                       d_grad_absmax_.ptr);
    quant_seed_ = quant_seed_ * 1664525u + 1013904223u;
    hipLaunchKernelGGL(hipk::k_grad_quantize, dim3((num_data_ + 255) / 256), dim3(256), 0,
                       stream_, GradPtr(), HessPtr(), num_data_, d_grad_absmax_.ptr,
                       quant_levels_, config_->stochastic_rounding ? 1 : 0, quant_seed_,
                       d_grad_packed_.ptr, d_grad_scales_.ptr);
  }
  // root setup + stats (global count via RCCL when distributed)
  hipLaunchKernelGGL(hipk::k_init_root, dim3(1), dim3(1), 0, stream_, d_leaf_begin_.ptr,
                     d_leaf_cnt_.ptr, d_leaf_slot_.ptr, d_leaf_stats_.ptr,
                     static_cast<int>(used_cnt_), d_gbuf_.ptr, d_counters_.ptr,
                     d_root_leaf_.ptr, d_minus1_.ptr, d_leaf_bounds_.ptr,
                     d_leaf_branch_.ptr);
  {
    const int blocks = std::min(2048, (static_cast<int>(used_cnt_) + 255) / 256);
    hipLaunchKernelGGL(hipk::k_root_sums, dim3(blocks), dim3(256), 0, stream_, d_idx_.ptr,
                       static_cast<int>(used_cnt_), GradPtr(), HessPtr(),
                       d_leaf_stats_.ptr);
  }
  auto& comm = Comm();
  dist_ = comm.active();
  own_scan_ = false;
  if (dist_ && comm.World() > 1) {
    // comm-path selection: full-histogram allreduce for small
    // payloads (latency-bound over xGMI), reduce-scatter + per-rank feature
    // ownership for large ones (divides wire bytes AND the gain scan by world).
    // MIGBM_DIST_HIST=allreduce|reduce_scatter overrides the size heuristic.
    const char* e = getenv("MIGBM_DIST_HIST");
    const std::string mode = e ? e : "auto";
    const size_t payload = static_cast<size_t>(total_bins_) * 2 * sizeof(float);
    const bool want_rs =
        mode == "reduce_scatter" || (mode == "auto" && payload >= (128u << 10));
    if (want_rs) own_scan_ = SetupOwnership(comm.World(), comm.Rank());
  }
  // forced splits: the schedule is data-independent (validity checks are all
  // host-side), so simulate the CPU loop's leaf bookkeeping up front
  forced_sched_.clear();
  if (forced_root2_) {
    own_scan_ = false;  // forced features may not be rank-owned in rs mode
    std::vector<const ForcedNode*> fol(nl, nullptr);
    fol[0] = forced_root2_.get();
    int sim_leaves = 1;
    for (int it = 0; it < nl - 1; ++it) {
      ForcedStep step;
      for (int l = 0; l < sim_leaves; ++l) {
        const ForcedNode* node = fol[l];
        if (node == nullptr) continue;
        const int inner = train_data_->InnerFeatureIndex(node->feature);
        const BinMapper* mp = inner >= 0 ? train_data_->FeatureBinMapper(inner) : nullptr;
        int bin = -1;
        if (mp != nullptr && mp->bin_type() == BinType::kNumerical)
          bin = std::min<int>(static_cast<int>(mp->ValueToBin(node->threshold)),
                              mp->num_numeric_bin() - 2);
        if (bin < 0) {
          fol[l] = nullptr;  // unusable forced node: cleared, scan continues
          continue;
        }
        step = {l, inner, bin};
        fol[l] = node->left.get();
        fol[sim_leaves] = node->right.get();
        break;
      }
      forced_sched_.push_back(step);
      ++sim_leaves;
      if (step.leaf < 0) {
        bool any = false;
        for (int l = 0; l < sim_leaves; ++l) any = any || fol[l] != nullptr;
        if (!any) break;  // no pending forced nodes: normal loop from here on
      }
    }
    while (!forced_sched_.empty() && forced_sched_.back().leaf < 0)
      forced_sched_.pop_back();
  }
  if (dist_) {
    comm.AllReduce(&d_leaf_stats_.ptr[0].sum_g, 2, stream_);
    comm.AllReduce(d_gbuf_.ptr, 1, stream_);
  }
  {
    // global root count + the root's own output (the smoothing parent for depth-1
    // candidates); single-GPU gbuf already holds the local count
    hipk::GainParams rp = {};
    rp.l1 = config_->lambda_l1;
    rp.l2 = config_->lambda_l2;
    rp.mds = config_->max_delta_step;
    hipLaunchKernelGGL(hipk::k_set_root_global_cnt, dim3(1), dim3(1), 0, stream_,
                       d_leaf_stats_.ptr, d_gbuf_.ptr, rp);
  }

  // root histogram + best split (leafB disabled via literal -1 pointer semantics)
  LaunchHist(d_root_leaf_.ptr, 0, HistBlocksFor(static_cast<int>(used_cnt_)));
  ReduceSpareHist(0);
  LaunchBestSplit(d_root_leaf_.ptr, 0);
  if (own_scan_) SyncGlobalWinner();

  // ---- device-driven split loop: ZERO host syncs; winner decisions accumulate in
  // d_split_log_ and the tree is replayed on the host after one end-of-tree download.
  const int kPartBlocks = 256;
  const int kLoopHistBlocks = 768;
  for (int split_i = 0; split_i < nl - 1; ++split_i) {
    // winner for this split was already selected by the fused k_best_leaf_overall
    // at the end of the previous split (or of the root best-split pass)
    if (split_i < static_cast<int>(forced_sched_.size()) &&
        forced_sched_[split_i].leaf >= 0) {
      LaunchForcedWinner(forced_sched_[split_i]);  // forced splits take priority
    }
    if (coop_launch_) {
      void* args[] = {
          const_cast<void*>(static_cast<const void*>(&d_idx_.ptr)),
          static_cast<void*>(&d_idx_tmp_.ptr),
          static_cast<void*>(&d_leaf_begin_.ptr),
          static_cast<void*>(&d_leaf_cnt_.ptr),
          static_cast<void*>(&d_winner_leaf_.ptr),
          static_cast<void*>(&d_winner_.ptr),
          static_cast<void*>(&d_feat_meta_.ptr),
          static_cast<void*>(&d_cols_.ptr),
          static_cast<void*>(&num_data_),
          static_cast<void*>(&d_marks_.ptr),
          static_cast<void*>(&d_block_cnt_.ptr),
          static_cast<void*>(&d_block_loff_.ptr),
          static_cast<void*>(&d_block_roff_.ptr),
          static_cast<void*>(&d_ctr_.ptr),
          static_cast<void*>(&d_gbuf_.ptr),
          static_cast<void*>(&d_idx_.ptr),
      };
      HIP_OK(hipLaunchCooperativeKernel(
          reinterpret_cast<void*>(hipk::k_part_fused), dim3(kPartBlocks),
          dim3(kHistBlock), args, 0, stream_));
    } else {
      if (rows16_) {
        hipLaunchKernelGGL((hipk::k_part_mark<uint16_t>), dim3(kPartBlocks),
                           dim3(kHistBlock), 0, stream_, d_idx_.ptr, d_leaf_begin_.ptr,
                           d_leaf_cnt_.ptr, d_winner_leaf_.ptr, d_winner_.ptr,
                           d_feat_meta_.ptr,
                           reinterpret_cast<const uint16_t*>(d_cols_.ptr), num_data_,
                           d_marks_.ptr, d_block_cnt_.ptr);
      } else {
        hipLaunchKernelGGL((hipk::k_part_mark<uint8_t>), dim3(kPartBlocks),
                           dim3(kHistBlock), 0, stream_, d_idx_.ptr, d_leaf_begin_.ptr,
                           d_leaf_cnt_.ptr, d_winner_leaf_.ptr, d_winner_.ptr,
                           d_feat_meta_.ptr, d_cols_.ptr, num_data_, d_marks_.ptr,
                           d_block_cnt_.ptr);
      }
      hipLaunchKernelGGL(hipk::k_part_scan, dim3(1), dim3(256), 0, stream_, d_block_cnt_.ptr,
                         kPartBlocks, d_leaf_cnt_.ptr, d_winner_leaf_.ptr, d_block_loff_.ptr,
                         d_block_roff_.ptr, d_ctr_.ptr);
      hipLaunchKernelGGL(hipk::k_part_scatter, dim3(kPartBlocks), dim3(kHistBlock), 0,
                         stream_, d_idx_.ptr, d_idx_tmp_.ptr, d_leaf_begin_.ptr,
                         d_leaf_cnt_.ptr, d_winner_leaf_.ptr, d_marks_.ptr,
                         d_block_loff_.ptr, d_block_roff_.ptr, d_ctr_.ptr);
      hipLaunchKernelGGL(hipk::k_copy_back, dim3(kPartBlocks), dim3(kHistBlock), 0, stream_,
                         d_idx_tmp_.ptr, d_idx_.ptr, d_leaf_begin_.ptr, d_leaf_cnt_.ptr,
                         d_winner_leaf_.ptr, d_ctr_.ptr, d_gbuf_.ptr);
    }
    if (dist_) comm.AllReduce(d_gbuf_.ptr, 1, stream_);
    if (hist_dp_)
      hipLaunchKernelGGL(hipk::k_finalize, dim3(1), dim3(256), 0, stream_,
                         d_leaf_begin_.ptr, d_leaf_cnt_.ptr, d_leaf_slot_.ptr,
                         d_leaf_stats_.ptr, d_winner_.ptr, d_winner_leaf_.ptr,
                         d_counters_.ptr, d_split_log_.ptr, d_ctr_.ptr, d_gbuf_.ptr,
                         reinterpret_cast<double*>(d_hist_.ptr),
                         static_cast<size_t>(total_bins_) * 2, total_bins_ * 2,
                         use_cegb_ ? d_cegb_coupled_.ptr : nullptr,
                         use_mono_ ? d_mono_.ptr : nullptr, d_leaf_bounds_.ptr,
                         d_leaf_branch_.ptr);
    else
      hipLaunchKernelGGL(hipk::k_finalize, dim3(1), dim3(256), 0, stream_,
                         d_leaf_begin_.ptr, d_leaf_cnt_.ptr, d_leaf_slot_.ptr,
                         d_leaf_stats_.ptr, d_winner_.ptr, d_winner_leaf_.ptr,
                         d_counters_.ptr, d_split_log_.ptr, d_ctr_.ptr, d_gbuf_.ptr,
                         d_hist_.ptr, static_cast<size_t>(total_bins_) * 2,
                         total_bins_ * 2, use_cegb_ ? d_cegb_coupled_.ptr : nullptr,
                         use_mono_ ? d_mono_.ptr : nullptr,
                         d_leaf_bounds_.ptr, d_leaf_branch_.ptr);
    LaunchHist(d_winner_leaf_.ptr, 1, kLoopHistBlocks, /*zero_spare=*/false);
    ReduceSpareHist(split_i + 1);  // spare slot for split i is deterministically i+1
    {
      const int n_elem = total_bins_ * 2;
      if (hist_dp_)
        hipLaunchKernelGGL(hipk::k_hist_subtract, dim3((n_elem + 1023) / 1024), dim3(256),
                           0, stream_, reinterpret_cast<double*>(d_hist_.ptr),
                           static_cast<size_t>(total_bins_) * 2, d_leaf_slot_.ptr,
                           d_leaf_stats_.ptr, d_winner_leaf_.ptr, d_counters_.ptr, n_elem);
      else
        hipLaunchKernelGGL(hipk::k_hist_subtract, dim3((n_elem + 1023) / 1024), dim3(256),
                           0, stream_, d_hist_.ptr, static_cast<size_t>(total_bins_) * 2,
                           d_leaf_slot_.ptr, d_leaf_stats_.ptr, d_winner_leaf_.ptr,
                           d_counters_.ptr, n_elem);
    }
    LaunchBestSplit(d_winner_leaf_.ptr, 1);
    if (own_scan_) SyncGlobalWinner();
  }

  // ---- one download: split log + exact leaf layout; replay the tree on the host
  host_log_.resize(nl);
  HIP_OK(hipMemcpyAsync(host_log_.data(), d_split_log_.ptr, sizeof(hipk::LogEntry) * nl,
                        hipMemcpyDeviceToHost, stream_));
  HIP_OK(hipMemcpyAsync(leaf_begin_.data(), d_leaf_begin_.ptr, sizeof(int) * nl,
                        hipMemcpyDeviceToHost, stream_));
  HIP_OK(hipMemcpyAsync(leaf_cnt_.data(), d_leaf_cnt_.ptr, sizeof(int) * nl,
                        hipMemcpyDeviceToHost, stream_));
  HIP_OK(hipStreamSynchronize(stream_));

  for (int i = 0; i < nl - 1; ++i) {
    const hipk::LogEntry& e = host_log_[i];
    if (e.leaf < 0) break;
    const hipk::SplitRec& w = e.rec;
    const int L = e.leaf;
    const int f = w.feature;
    const BinMapper* mapper = train_data_->FeatureBinMapper(f);
    const int orig_f = train_data_->RealFeatureIndex(f);
    if (mapper->bin_type() == BinType::kCategorical) {
      // value-level bitset from the winner: 64-bit bin subset (sorted scan) or a
      // single one-hot bin
      std::vector<int> cats;
      const bool has_mask =
          (w.cat_mask[0] | w.cat_mask[1] | w.cat_mask[2] | w.cat_mask[3]) != 0ull;
      if (has_mask) {
        for (int b = 0; b < 256; ++b)
          if ((w.cat_mask[b >> 6] >> (b & 63)) & 1ull)
            cats.push_back(static_cast<int>(mapper->BinToValue(b)));
      } else {
        cats.push_back(static_cast<int>(mapper->BinToValue(w.bin)));
      }
      int max_cat = 0;
      for (int c : cats) max_cat = std::max(max_cat, c);
      std::vector<uint32_t> bits(max_cat / 32 + 1, 0);
      for (int c : cats)
        if (c >= 0) bits[c >> 5] |= 1u << (c & 31);
      tree->SplitCategorical(L, f, orig_f, bits.data(), static_cast<int>(bits.size()),
                             w.left_out, w.right_out, w.left_cnt, w.right_cnt, w.left_h,
                             0.0, static_cast<float>(w.gain), mapper->missing_type());
    } else {
      tree->Split(L, f, orig_f, w.bin, mapper->BinToValue(w.bin), w.left_out, w.right_out,
                  w.left_cnt, w.right_cnt, w.left_h, 0.0, static_cast<float>(w.gain),
                  mapper->missing_type(), w.default_left != 0);
    }
  }
  // fix the tree's leaf counts with exact values (internal counts recomputed
  // inside). Distributed: LeafStat.cnt carries the GLOBAL count so every rank's
  // model (and its leaf_count fields) is byte-identical.
  {
    std::vector<int> counts(tree->num_leaves());
    if (dist_) {
      std::vector<hipk::LeafStat> st(tree->num_leaves());
      HIP_OK(hipMemcpy(st.data(), d_leaf_stats_.ptr,
                       sizeof(hipk::LeafStat) * tree->num_leaves(),
                       hipMemcpyDeviceToHost));
      for (int l = 0; l < tree->num_leaves(); ++l) counts[l] = st[l].cnt;
    } else {
      for (int l = 0; l < tree->num_leaves(); ++l) counts[l] = leaf_cnt_[l];
    }
    tree->OverrideLeafCounts(counts);
  }
  if (linear_) CalculateLinearDevice(tree.get());
  return tree.release();
}

void HIPTreeLearner::AddPredictionToScore(const Tree* tree, double* /*out_score*/) {
  const int nl = tree->num_leaves();
  std::vector<int> order(nl);
  for (int l = 0; l < nl; ++l) order[l] = l;
  std::sort(order.begin(), order.end(),
            [&](int a, int b) { return leaf_begin_[a] < leaf_begin_[b]; });
  std::vector<double> outs(nl);
  std::vector<int> sorted_begin(nl);
  for (int k = 0; k < nl; ++k) {
    outs[k] = tree->LeafOutput(order[k]);
    sorted_begin[k] = leaf_begin_[order[k]];
  }
  HIP_OK(hipMemcpyAsync(d_leaf_out_.ptr, outs.data(), sizeof(double) * nl,
                        hipMemcpyHostToDevice, stream_));
  HIP_OK(hipMemcpyAsync(d_sorted_begin_.ptr, sorted_begin.data(), sizeof(int) * nl,
                        hipMemcpyHostToDevice, stream_));
  const int n = static_cast<int>(used_cnt_);
  if (tree->is_linear()) {
    // linear leaves: score += const + coeff . raw (sorted-leaf-order flattening)
    std::vector<double> lin_const(nl), coeff_flat;
    std::vector<int> coeff_off(nl), coeff_cnt(nl), feat_flat;
    for (int k2 = 0; k2 < nl; ++k2) {
      const int l = order[k2];
      coeff_off[k2] = static_cast<int>(feat_flat.size());
      const auto& fi = tree->leaf_features_inner(l);
      const auto& co = tree->leaf_coeffs(l);
      coeff_cnt[k2] = static_cast<int>(co.size());
      lin_const[k2] = co.empty() ? tree->LeafOutput(l) : tree->leaf_const(l);
      for (size_t j = 0; j < co.size(); ++j) {
        feat_flat.push_back(fi[j]);
        coeff_flat.push_back(co[j]);
      }
    }
    if (feat_flat.empty()) { feat_flat.push_back(0); coeff_flat.push_back(0.0); }
    d_lin_const_.Alloc(nl);
    d_lin_coeff_off_.Alloc(nl);
    d_lin_coeff_cnt_.Alloc(nl);
    d_lin_feat_flat_.Alloc(feat_flat.size());
    d_lin_coeff_flat_.Alloc(coeff_flat.size());
    HIP_OK(hipMemcpyAsync(d_lin_const_.ptr, lin_const.data(), sizeof(double) * nl,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_lin_coeff_off_.ptr, coeff_off.data(), sizeof(int) * nl,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_lin_coeff_cnt_.ptr, coeff_cnt.data(), sizeof(int) * nl,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_lin_feat_flat_.ptr, feat_flat.data(),
                          sizeof(int) * feat_flat.size(), hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_lin_coeff_flat_.ptr, coeff_flat.data(),
                          sizeof(double) * coeff_flat.size(), hipMemcpyHostToDevice,
                          stream_));
    hipLaunchKernelGGL(hipk::k_score_update_linear, dim3((n + 255) / 256), dim3(256), 0,
                       stream_, d_idx_.ptr, d_sorted_begin_.ptr, nl, n, d_lin_const_.ptr,
                       d_leaf_out_.ptr, d_lin_coeff_off_.ptr, d_lin_coeff_cnt_.ptr,
                       d_lin_feat_flat_.ptr, d_lin_coeff_flat_.ptr, d_raw_.ptr, num_data_,
                       ScorePtr());
  } else {
  hipLaunchKernelGGL(hipk::k_score_update, dim3((n + 255) / 256), dim3(256), 0, stream_,
                     d_idx_.ptr, d_sorted_begin_.ptr, nl, n, d_leaf_out_.ptr, ScorePtr());
  }
  if (bag_indices_ != nullptr && bag_cnt_ > 0 &&
      bag_cnt_ < static_cast<data_size_t>(num_data_)) {
    std::vector<uint32_t> oob;
    oob.reserve(num_data_ - bag_cnt_);
    size_t bi = 0;
    for (int i = 0; i < num_data_; ++i) {
      if (bi < bag_cnt_ && bag_indices_[bi] == i) ++bi;
      else oob.push_back(i);
    }
    const int ni = nl - 1;
    if (ni > 0 && !oob.empty()) {
      std::vector<int> feat(ni), thr(ni), lc(ni), rc(ni), nb(ni);
      std::vector<uint8_t> dl(ni), iscat(ni, 0);
      std::vector<unsigned long long> catmask;
      bool any_cat = false;
      for (int i2 = 0; i2 < ni; ++i2) {
        feat[i2] = tree->split_feature_inner(i2);
        thr[i2] = static_cast<int>(tree->threshold_in_bin(i2));
        lc[i2] = tree->left_child(i2);
        rc[i2] = tree->right_child(i2);
        nb[i2] = feat_meta_host_[feat[i2]].nan_bin;
        dl[i2] = (tree->decision_type(i2) & Tree::kDefaultLeftMask) ? 1 : 0;
        iscat[i2] = tree->IsCategoricalSplit(i2) ? 1 : 0;
        any_cat = any_cat || iscat[i2];
      }
      if (any_cat) {
        // category-space node bitsets -> bin-space 4-word masks for the walk
        catmask.assign(static_cast<size_t>(4) * ni, 0ull);
        const auto& cb = tree->cat_boundaries();
        const auto& ct = tree->cat_threshold();
        for (int i2 = 0; i2 < ni; ++i2) {
          if (!iscat[i2]) continue;
          const int cat_idx = thr[i2];
          const uint32_t* bits = ct.data() + cb[cat_idx];
          const int n_words = cb[cat_idx + 1] - cb[cat_idx];
          const BinMapper* m = train_data_->FeatureBinMapper(feat[i2]);
          const int nbin = std::min(m->num_bin(), 256);
          for (int b = 0; b < nbin; ++b) {
            const int cat = static_cast<int>(m->BinToValue(b));
            if (cat >= 0 && (cat >> 5) < n_words && ((bits[cat >> 5] >> (cat & 31)) & 1))
              catmask[4 * i2 + (b >> 6)] |= 1ull << (b & 63);
          }
        }
      }
      // linear leaves: coefficient arrays in TREE-leaf order for the walk
      std::vector<double> lc_const, lc_flat;
      std::vector<int> lc_off, lc_cnt, lf_flat;
      if (tree->is_linear()) {
        lc_const.resize(nl);
        lc_off.resize(nl);
        lc_cnt.resize(nl);
        for (int l = 0; l < nl; ++l) {
          lc_off[l] = static_cast<int>(lf_flat.size());
          const auto& fi = tree->leaf_features_inner(l);
          const auto& co = tree->leaf_coeffs(l);
          lc_cnt[l] = static_cast<int>(co.size());
          lc_const[l] = co.empty() ? tree->LeafOutput(l) : tree->leaf_const(l);
          for (size_t j = 0; j < co.size(); ++j) {
            lf_flat.push_back(fi[j]);
            lc_flat.push_back(co[j]);
          }
        }
        if (lf_flat.empty()) { lf_flat.push_back(0); lc_flat.push_back(0.0); }
      }
      std::vector<double> out_by_leaf(nl);
      for (int l = 0; l < nl; ++l) out_by_leaf[l] = tree->LeafOutput(l);
      HIP_OK(hipMemcpyAsync(d_tw_feat_.ptr, feat.data(), sizeof(int) * ni,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_thr_.ptr, thr.data(), sizeof(int) * ni,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_left_.ptr, lc.data(), sizeof(int) * ni,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_right_.ptr, rc.data(), sizeof(int) * ni,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_nan_.ptr, nb.data(), sizeof(int) * ni,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_dl_.ptr, dl.data(), ni, hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_tw_out_.ptr, out_by_leaf.data(), sizeof(double) * nl,
                            hipMemcpyHostToDevice, stream_));
      HIP_OK(hipMemcpyAsync(d_oob_.ptr, oob.data(), sizeof(uint32_t) * oob.size(),
                            hipMemcpyHostToDevice, stream_));
      const unsigned long long* catp = nullptr;
      const uint8_t* iscatp = nullptr;
      if (any_cat) {
        d_tw_iscat_.Alloc(ni);
        d_tw_cat_.Alloc(catmask.size());
        HIP_OK(hipMemcpyAsync(d_tw_iscat_.ptr, iscat.data(), ni, hipMemcpyHostToDevice,
                              stream_));
        HIP_OK(hipMemcpyAsync(d_tw_cat_.ptr, catmask.data(),
                              sizeof(unsigned long long) * catmask.size(),
                              hipMemcpyHostToDevice, stream_));
        catp = d_tw_cat_.ptr;
        iscatp = d_tw_iscat_.ptr;
      }
      const int *loffp = nullptr, *lcntp = nullptr, *lfeatp = nullptr;
      const double *lcoefp = nullptr, *lconstp = nullptr;
      if (tree->is_linear()) {
        d_twl_off_.Alloc(nl);
        d_twl_cnt_.Alloc(nl);
        d_twl_const_.Alloc(nl);
        d_twl_feat_.Alloc(lf_flat.size());
        d_twl_coeff_.Alloc(lc_flat.size());
        HIP_OK(hipMemcpyAsync(d_twl_off_.ptr, lc_off.data(), sizeof(int) * nl,
                              hipMemcpyHostToDevice, stream_));
        HIP_OK(hipMemcpyAsync(d_twl_cnt_.ptr, lc_cnt.data(), sizeof(int) * nl,
                              hipMemcpyHostToDevice, stream_));
        HIP_OK(hipMemcpyAsync(d_twl_const_.ptr, lc_const.data(), sizeof(double) * nl,
                              hipMemcpyHostToDevice, stream_));
        HIP_OK(hipMemcpyAsync(d_twl_feat_.ptr, lf_flat.data(), sizeof(int) * lf_flat.size(),
                              hipMemcpyHostToDevice, stream_));
        HIP_OK(hipMemcpyAsync(d_twl_coeff_.ptr, lc_flat.data(),
                              sizeof(double) * lc_flat.size(), hipMemcpyHostToDevice,
                              stream_));
        loffp = d_twl_off_.ptr;
        lcntp = d_twl_cnt_.ptr;
        lfeatp = d_twl_feat_.ptr;
        lcoefp = d_twl_coeff_.ptr;
        lconstp = d_twl_const_.ptr;
      }
      if (rows16_) {
        hipLaunchKernelGGL((hipk::k_tree_predict_add<uint16_t>),
                           dim3((static_cast<int>(oob.size()) + 255) / 256), dim3(256), 0,
                           stream_, reinterpret_cast<const uint16_t*>(d_cols_.ptr),
                           num_data_, d_tw_feat_.ptr, d_tw_thr_.ptr, d_tw_left_.ptr,
                           d_tw_right_.ptr, d_tw_nan_.ptr, d_tw_dl_.ptr, iscatp, catp,
                           d_tw_out_.ptr, loffp, lcntp, lfeatp, lcoefp, lconstp,
                           linear_ ? d_raw_.ptr : nullptr, d_oob_.ptr,
                           static_cast<int>(oob.size()), ScorePtr());
      } else {
        hipLaunchKernelGGL((hipk::k_tree_predict_add<uint8_t>),
                           dim3((static_cast<int>(oob.size()) + 255) / 256), dim3(256), 0,
                           stream_, d_cols_.ptr, num_data_, d_tw_feat_.ptr, d_tw_thr_.ptr,
                           d_tw_left_.ptr, d_tw_right_.ptr, d_tw_nan_.ptr, d_tw_dl_.ptr,
                           iscatp, catp, d_tw_out_.ptr, loffp, lcntp, lfeatp, lcoefp,
                           lconstp, linear_ ? d_raw_.ptr : nullptr, d_oob_.ptr,
                           static_cast<int>(oob.size()), ScorePtr());
      }
    }
  }
}

void HIPTreeLearner::CalculateLinearDevice(Tree* tree) {
  tree->SetLinear(true);
  const int nl = tree->num_leaves();
  if (nl <= 1) return;
  // branch-path features per leaf (numerical, unique, inner ids) — host walk
  std::vector<std::vector<int>> leaf_feats(nl);
  std::function<void(int, std::vector<int>&)> walk = [&](int node, std::vector<int>& path) {
    if (node < 0) {
      leaf_feats[~node] = path;
      return;
    }
    const int fi = tree->split_feature_inner(node);
    bool added = false;
    if (!tree->IsCategoricalSplit(node) &&
        std::find(path.begin(), path.end(), fi) == path.end()) {
      path.push_back(fi);
      added = true;
    }
    walk(tree->left_child(node), path);
    walk(tree->right_child(node), path);
    if (added) path.pop_back();
  };
  std::vector<int> path;
  walk(0, path);

  constexpr int kMaxK = 32;  // LDS/register cap; deeper unique paths stay constant
  // device Gram accumulation per leaf, then tiny host Cholesky solves
  for (int l = 0; l < nl; ++l) {
    const auto& feats = leaf_feats[l];
    const int k = static_cast<int>(feats.size());
    if (k == 0 || k > kMaxK || tree->leaf_count(l) < k + 2) continue;
    const int dim = k + 1;
    d_lin_feats_.Alloc(kMaxK);
    d_lin_A_.Alloc((kMaxK + 1) * (kMaxK + 1));
    d_lin_b_.Alloc(kMaxK + 1);
    if (!d_lin_nan_.ptr) d_lin_nan_.Alloc(1);
    HIP_OK(hipMemcpyAsync(d_lin_feats_.ptr, feats.data(), sizeof(int) * k,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemsetAsync(d_lin_A_.ptr, 0, sizeof(double) * dim * dim, stream_));
    HIP_OK(hipMemsetAsync(d_lin_b_.ptr, 0, sizeof(double) * dim, stream_));
    HIP_OK(hipMemsetAsync(d_lin_nan_.ptr, 0, sizeof(int), stream_));
    const int blocks = std::min(64, (leaf_cnt_[l] + 255) / 256);
    const size_t lds = sizeof(double) * (dim * dim + dim);
    hipLaunchKernelGGL(hipk::k_linear_gram, dim3(blocks), dim3(256), lds, stream_,
                       d_idx_.ptr, leaf_begin_[l], leaf_cnt_[l], d_raw_.ptr, num_data_,
                       d_lin_feats_.ptr, k, GradPtr(), HessPtr(), d_lin_A_.ptr,
                       d_lin_b_.ptr, d_lin_nan_.ptr);
    if (dist_ && Comm().World() > 1) {
      // shard Gram matrices sum to the global fit (all ranks end identical)
      Comm().AllReduce(d_lin_A_.ptr, static_cast<size_t>(dim) * dim, stream_);
      Comm().AllReduce(d_lin_b_.ptr, static_cast<size_t>(dim), stream_);
    }
    std::vector<double> A(dim * dim), b(dim);
    int nan_flag = 0;
    HIP_OK(hipMemcpyAsync(A.data(), d_lin_A_.ptr, sizeof(double) * dim * dim,
                          hipMemcpyDeviceToHost, stream_));
    HIP_OK(hipMemcpyAsync(b.data(), d_lin_b_.ptr, sizeof(double) * dim,
                          hipMemcpyDeviceToHost, stream_));
    HIP_OK(hipMemcpyAsync(&nan_flag, d_lin_nan_.ptr, sizeof(int), hipMemcpyDeviceToHost,
                          stream_));
    HIP_OK(hipStreamSynchronize(stream_));
    if (dist_ && Comm().World() > 1) {
      // a NaN on ANY rank must disable the leaf's linear fit everywhere
      double nf2 = nan_flag;
      DevBuf<double> scratch;
      scratch.Alloc(1);
      HIP_OK(hipMemcpy(scratch.ptr, &nf2, sizeof(double), hipMemcpyHostToDevice));
      Comm().AllReduce(scratch.ptr, 1, stream_);
      HIP_OK(hipStreamSynchronize(stream_));
      HIP_OK(hipMemcpy(&nf2, scratch.ptr, sizeof(double), hipMemcpyDeviceToHost));
      nan_flag = nf2 > 0 ? 1 : 0;
    }
    if (nan_flag) continue;  // leaves with missing path values stay constant
    for (int a = 0; a < k; ++a)
      A[a * dim + a] += config_->linear_lambda + config_->lambda_l2;
    A[k * dim + k] += config_->lambda_l2;
    std::vector<double> beta;
    if (!CholeskySolve(A, b, dim, &beta)) continue;
    std::vector<int> feats_real(k);
    for (int j = 0; j < k; ++j) feats_real[j] = train_data_->RealFeatureIndex(feats[j]);
    std::vector<double> coeffs(beta.begin(), beta.begin() + k);
    tree->SetLeafLinear(l, beta[k], feats_real, feats, coeffs);
  }
}

void HIPTreeLearner::RenewTreeOutput(Tree* tree, const ObjectiveFunction* obj,
                                     std::function<double(const label_t*, int)>, data_size_t,
                                     const data_size_t*, data_size_t,
                                     const double* train_score) {
  if (obj == nullptr || !obj->NeedRenewTreeOutput()) return;
  (void)train_score;
  const std::string name = obj->GetName();
  const int nl = tree->num_leaves();
  double alpha = 0.5;
  int wmode = -1;
  if (name == "regression_l1") wmode = weights_present_ ? 1 : 0;
  else if (name == "quantile") { alpha = config_->alpha; wmode = weights_present_ ? 1 : 0; }
  else if (name == "mape") wmode = 2;
  if (wmode >= 0 && !config_->reg_sqrt) {
    // device percentile renewal: no score download, no host sort
    std::vector<double> outs(nl);
    for (int l = 0; l < nl; ++l) outs[l] = tree->LeafOutput(l);
    HIP_OK(hipMemcpyAsync(d_leaf_out_.ptr, outs.data(), sizeof(double) * nl,
                          hipMemcpyHostToDevice, stream_));
    hipLaunchKernelGGL(hipk::k_renew_percentile, dim3(nl), dim3(256), 0, stream_,
                       d_idx_.ptr, d_leaf_begin_.ptr, d_leaf_cnt_.ptr, nl, ScorePtr(),
                       d_label_.ptr, weights_present_ ? d_weight_.ptr : nullptr, wmode,
                       alpha, d_leaf_out_.ptr);
    HIP_OK(hipMemcpyAsync(outs.data(), d_leaf_out_.ptr, sizeof(double) * nl,
                          hipMemcpyDeviceToHost, stream_));
    HIP_OK(hipStreamSynchronize(stream_));
    if (dist_ && Comm().World() > 1) {
      // ranks renewed from their local shard: sync to the count-weighted mean so
      // every rank keeps the identical model (the reference's multi-GPU mode
      // skips renewal entirely, nccl_gbdt.cpp:167-171 — this is strictly closer)
      std::vector<double> acc(2 * nl);
      for (int l = 0; l < nl; ++l) {
        acc[2 * l] = outs[l] * leaf_cnt_[l];
        acc[2 * l + 1] = leaf_cnt_[l];
      }
      DevBuf<double> scratch;
      scratch.Alloc(2 * nl);
      HIP_OK(hipMemcpyAsync(scratch.ptr, acc.data(), sizeof(double) * 2 * nl,
                            hipMemcpyHostToDevice, stream_));
      Comm().AllReduce(scratch.ptr, 2 * nl, stream_);
      HIP_OK(hipMemcpyAsync(acc.data(), scratch.ptr, sizeof(double) * 2 * nl,
                            hipMemcpyDeviceToHost, stream_));
      HIP_OK(hipStreamSynchronize(stream_));
      for (int l = 0; l < nl; ++l)
        if (acc[2 * l + 1] > 0) outs[l] = acc[2 * l] / acc[2 * l + 1];
    }
    for (int l = 0; l < nl; ++l)
      if (leaf_cnt_[l] > 0 || (dist_ && outs[l] != tree->LeafOutput(l)))
        tree->SetLeafOutput(l, outs[l]);
    return;
  }
  // host fallback (unknown renewing objective)
  std::vector<uint32_t> idx(used_cnt_);
  HIP_OK(hipStreamSynchronize(stream_));
  HIP_OK(hipMemcpy(idx.data(), d_idx_.ptr, sizeof(uint32_t) * used_cnt_,
                   hipMemcpyDeviceToHost));
  std::vector<double> score(num_data_);
  HIP_OK(hipMemcpy(score.data(), ScorePtr(), sizeof(double) * num_data_,
                   hipMemcpyDeviceToHost));
  for (int l = 0; l < nl; ++l) {
    if (leaf_cnt_[l] == 0) continue;
    std::vector<data_size_t> rows(leaf_cnt_[l]);
    for (int i = 0; i < leaf_cnt_[l]; ++i)
      rows[i] = static_cast<data_size_t>(idx[leaf_begin_[l] + i]);
    tree->SetLeafOutput(l, obj->RenewTreeOutput(tree->LeafOutput(l), rows.data(),
                                                leaf_cnt_[l], score.data()));
  }
}

bool HIPTreeLearner::DeviceEvalPointwise(int loss_kind, double loss_a, int convert_kind,
                                         double convert_param, double* out_sum,
                                         double* out_wsum) {
  if (loss_kind < 0 || num_class_score_ > 1) return false;
  if (!d_eval_out_.ptr) d_eval_out_.Alloc(2);
  HIP_OK(hipMemsetAsync(d_eval_out_.ptr, 0, 2 * sizeof(double), stream_));
  const int blocks = std::min(1024, (num_data_ + 255) / 256);
  hipLaunchKernelGGL(hipk::k_metric_pointwise, dim3(blocks), dim3(256), 0, stream_,
                     loss_kind, loss_a, convert_kind, convert_param, ScorePtr(),
                     d_label_.ptr, weights_present_ ? d_weight_.ptr : nullptr, num_data_,
                     d_eval_out_.ptr);
  double host_out[2];
  HIP_OK(hipMemcpyAsync(host_out, d_eval_out_.ptr, 2 * sizeof(double),
                        hipMemcpyDeviceToHost, stream_));
  HIP_OK(hipStreamSynchronize(stream_));
  *out_sum = host_out[0];
  *out_wsum = host_out[1];
  return true;
}

double HIPTreeLearner::DebugRootHistMaxRelErr(const score_t* g, const score_t* h) {
  cur_class_ = 0;
  UploadGradients(g, h);
  used_cnt_ = num_data_;
  hipLaunchKernelGGL(hipk::k_iota, dim3((num_data_ + 255) / 256), dim3(256), 0, stream_,
                     d_idx_.ptr, num_data_);
  hipLaunchKernelGGL(hipk::k_init_root, dim3(1), dim3(1), 0, stream_, d_leaf_begin_.ptr,
                     d_leaf_cnt_.ptr, d_leaf_slot_.ptr, d_leaf_stats_.ptr,
                     static_cast<int>(used_cnt_), d_gbuf_.ptr, d_counters_.ptr,
                     d_root_leaf_.ptr, d_minus1_.ptr, d_leaf_bounds_.ptr,
                     d_leaf_branch_.ptr);
  {
    const int blocks = std::min(2048, (static_cast<int>(used_cnt_) + 255) / 256);
    hipLaunchKernelGGL(hipk::k_root_sums, dim3(blocks), dim3(256), 0, stream_, d_idx_.ptr,
                       static_cast<int>(used_cnt_), GradPtr(), HessPtr(),
                       d_leaf_stats_.ptr);
  }
  LaunchHist(d_root_leaf_.ptr, 0, HistBlocksFor(static_cast<int>(used_cnt_)));
  HIP_OK(hipStreamSynchronize(stream_));
  std::vector<double> dev(static_cast<size_t>(total_bins_) * 2);
  if (hist_dp_) {
    HIP_OK(hipMemcpy(dev.data(), d_hist_.ptr, sizeof(double) * dev.size(),
                     hipMemcpyDeviceToHost));
  } else {
    std::vector<float> devf(dev.size());
    HIP_OK(hipMemcpy(devf.data(), d_hist_.ptr, sizeof(float) * devf.size(),
                     hipMemcpyDeviceToHost));
    for (size_t i = 0; i < dev.size(); ++i) dev[i] = devf[i];
  }
  // fp64 host oracle over the identical rows/gradients
  std::vector<double> host(dev.size(), 0.0);
  std::vector<int8_t> used(train_data_->num_features(), 1);
  std::vector<data_size_t> idx(num_data_);
  for (int i = 0; i < num_data_; ++i) idx[i] = i;
  train_data_->ConstructHistograms(used, idx.data(), num_data_, g, h, host.data());
  double max_rel = 0.0;
  for (size_t i = 0; i < dev.size(); ++i) {
    const double hv = host[i];
    const double dv = dev[i];
    const double err = std::fabs(dv - hv) / std::max(1.0, std::fabs(hv));
    max_rel = std::max(max_rel, err);
  }
  return max_rel;
}

// ------------------------------------------------- single-process multi-GPU (num_gpu)
/*! persistent per-device worker threads: all HIP calls for shard r run on thread
 *  r (hipSetDevice is per-thread state). */
class GpuWorkers {
 public:
  explicit GpuWorkers(const std::vector<int>& devices) : n_(devices.size()) {
    done_.assign(n_, 0);
    for (size_t r = 0; r < n_; ++r) {
      threads_.emplace_back([this, r, dev = devices[r]] {
        HIP_OK(hipSetDevice(dev));
        uint64_t seen = 0;
        for (;;) {
          std::function<void(int)> job;
          {
            std::unique_lock<std::mutex> lk(mu_);
            cv_.wait(lk, [&] { return stop_ || job_id_ > seen; });
            if (stop_) return;
            seen = job_id_;
            job = *job_;
          }
          job(static_cast<int>(r));
          {
            std::lock_guard<std::mutex> lk(mu_);
            done_[r] = seen;
            done_cv_.notify_all();
          }
        }
      });
    }
  }
  ~GpuWorkers() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
      cv_.notify_all();
    }
    for (auto& t : threads_) t.join();
  }
  /*! run fn(shard) on every worker; rethrows the first failure */
  void RunAll(const std::function<void(int)>& fn) {
    std::exception_ptr err = nullptr;
    std::mutex err_mu;
    std::function<void(int)> wrapped = [&](int r) {
      try {
        fn(r);
      } catch (...) {
        std::lock_guard<std::mutex> lk(err_mu);
        if (!err) err = std::current_exception();
      }
    };
    uint64_t id;
    {
      std::lock_guard<std::mutex> lk(mu_);
      job_ = &wrapped;
      id = ++job_id_;
      cv_.notify_all();
    }
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [&] {
        for (size_t r = 0; r < n_; ++r)
          if (done_[r] < id) return false;
        return true;
      });
    }
    if (err) std::rethrow_exception(err);
  }

 private:
  size_t n_;
  std::vector<std::thread> threads_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  const std::function<void(int)>* job_ = nullptr;
  uint64_t job_id_ = 0;
  bool stop_ = false;
  std::vector<uint64_t> done_;
};

/*! Single-process multi-GPU data-parallel learner (`num_gpu=N`): rows are
 *  sharded contiguously over N device contexts, each owning a HIPTreeLearner on
 *  its own device + host thread; per-shard learners reduce histograms / winners
 *  through a private clique (ncclCommInitAll over distinct devices; in-process
 *  barrier transport when shards share one physical GPU, e.g. a 1-GPU test box)
 *  and every shard builds the identical tree. Train path needs NO torchrun
 *  rendezvous. Capability parity: reference NCCLGBDT/NCCLTopology
 *  (src/boosting/cuda/nccl_gbdt.cpp:84-209, cuda_nccl_topology.hpp:104-188) —
 *  improved: bagging and RenewTreeOutput work here (the reference skips both),
 *  and the per-shard learners inherit the reduce-scatter ownership mode. */
class MultiGpuTreeLearner : public TreeLearner {
 public:
  explicit MultiGpuTreeLearner(const Config* config) : config_(config) {}
  ~MultiGpuTreeLearner() override {
    if (workers_) {
      // learners own per-device resources: free them on their own threads
      workers_->RunAll([&](int r) { learners_[r].reset(); });
    }
    for (ncclComm_t c : rccl_comms_)
      if (c) ncclCommDestroy(c);
  }

  bool IsHIPLearner() const override { return true; }
  bool DeviceObjectiveSupported(const std::string& name) const override {
    if (config_->is_unbalance && name == "binary") return false;  // needs global counts
    if (name == "lambdarank") return false;  // queries are not shardable here
    HIPTreeLearner probe(config_);
    return probe.DeviceObjectiveSupported(name);
  }

  void Init(const Dataset* train_data, bool is_constant_hessian) override {
    train_data_ = train_data;
    is_constant_hessian_ = is_constant_hessian;
    num_data_ = train_data->num_data();
    if (train_data->metadata().query_boundaries() != nullptr)
      Log::Fatal("ranking objectives with num_gpu>1 in one process are not supported; "
                 "use one process per GPU (tree_learner=data)");
    for (ncclComm_t c : rccl_comms_)
      if (c) ncclCommDestroy(c);
    rccl_comms_.clear();
    int ndev = 0;
    HIP_OK(hipGetDeviceCount(&ndev));
    world_ = std::max(1, config_->num_gpu);
    if (num_data_ < world_ * 2) Log::Fatal("num_gpu=%d with only %d rows", world_, num_data_);
    std::vector<int> devices(world_);
    bool distinct = true;
    if (!config_->gpu_device_id_list.empty()) {
      // explicit device selection: "gpu_device_id_list=0,2,4,6"
      std::vector<int> ids;
      std::string tok;
      for (char ch : config_->gpu_device_id_list + ",") {
        if (ch == ',') {
          if (!tok.empty()) ids.push_back(atoi(tok.c_str()));
          tok.clear();
        } else {
          tok += ch;
        }
      }
      if (static_cast<int>(ids.size()) != world_)
        Log::Fatal("gpu_device_id_list has %d entries but num_gpu=%d",
                   static_cast<int>(ids.size()), world_);
      for (int r = 0; r < world_; ++r) {
        devices[r] = ids[r] % std::max(1, ndev);
        for (int q = 0; q < r; ++q) distinct = distinct && devices[q] != devices[r];
      }
    } else {
      for (int r = 0; r < world_; ++r) {
        devices[r] = r % std::max(1, ndev);
        distinct = distinct && devices[r] == r;
      }
    }
    if (!distinct)
      Log::Warning("num_gpu=%d > %d visible devices: shards share GPUs "
                   "(in-process clique transport; correctness mode)",
                   world_, ndev);
    // contiguous row shards
    shard_begin_.resize(world_ + 1);
    for (int r = 0; r <= world_; ++r)
      shard_begin_[r] = static_cast<data_size_t>(static_cast<int64_t>(num_data_) * r / world_);
    shards_.resize(world_);
    learners_.resize(world_);
    if (!workers_) workers_ = std::make_unique<GpuWorkers>(devices);
    // RCCL clique over distinct devices, host clique otherwise
    std::vector<ncclComm_t> comms(world_, nullptr);
    if (distinct && world_ > 1) {
      NCCL_OK(ncclCommInitAll(comms.data(), world_, devices.data()));
      rccl_comms_ = comms;
    } else if (world_ > 1) {
      clique_ = std::make_unique<InProcClique>(world_);
    }
    workers_->RunAll([&](int r) {
      std::vector<data_size_t> rows(shard_begin_[r + 1] - shard_begin_[r]);
      for (size_t i = 0; i < rows.size(); ++i)
        rows[i] = shard_begin_[r] + static_cast<data_size_t>(i);
      shards_[r] = train_data_->Subset(rows.data(), static_cast<data_size_t>(rows.size()));
      auto* l = new HIPTreeLearner(config_);
      if (world_ > 1) {
        if (!rccl_comms_.empty()) l->SetCommRccl(rccl_comms_[r], world_, r);
        else l->SetCommClique(clique_.get(), r);
      }
      l->Init(shards_[r].get(), is_constant_hessian);
      learners_[r].reset(l);
    });
  }
  void ResetTrainingData(const Dataset* train_data) override {
    Init(train_data, is_constant_hessian_);
  }
  void ResetConfig(const Config* config) override {
    config_ = config;
    workers_->RunAll([&](int r) { learners_[r]->ResetConfig(config); });
  }

  Tree* Train(const score_t* gradients, const score_t* hessians, bool is_first) override {
    std::vector<Tree*> trees(world_, nullptr);
    workers_->RunAll([&](int r) {
      // host-gradient fallback: each shard uploads its row slice
      trees[r] = learners_[r]->Train(gradients ? gradients + shard_begin_[r] : nullptr,
                                     hessians ? hessians + shard_begin_[r] : nullptr,
                                     is_first);
    });
    for (int r = 1; r < world_; ++r) delete trees[r];
    return trees[0];
  }

  void SetBaggingData(const Dataset* subset, const data_size_t* used_indices,
                      data_size_t num_data) override {
    if (subset != nullptr) Log::Fatal("HIP learner does not use dataset-subset bagging");
    if (used_indices == nullptr || num_data == 0 || num_data == num_data_) {
      workers_->RunAll([&](int r) { learners_[r]->SetBaggingData(nullptr, nullptr, 0); });
      return;
    }
    // split the (ascending) global bag indices into per-shard local index sets
    // — bagging works under num_gpu (the reference NCCLGBDT skips it entirely)
    bag_parts_.assign(world_, {});
    const data_size_t* end = used_indices + num_data;
    for (int r = 0; r < world_; ++r) {
      const data_size_t* lo = std::lower_bound(used_indices, end, shard_begin_[r]);
      const data_size_t* hi = std::lower_bound(used_indices, end, shard_begin_[r + 1]);
      bag_parts_[r].resize(hi - lo);
      for (size_t i = 0; i < bag_parts_[r].size(); ++i)
        bag_parts_[r][i] = lo[i] - shard_begin_[r];
    }
    workers_->RunAll([&](int r) {
      learners_[r]->SetBaggingData(nullptr, bag_parts_[r].data(),
                                   static_cast<data_size_t>(bag_parts_[r].size()));
    });
  }

  void AddPredictionToScore(const Tree* tree, double* out_score) override {
    workers_->RunAll([&](int r) { learners_[r]->AddPredictionToScore(tree, out_score); });
  }

  void RenewTreeOutput(Tree* tree, const ObjectiveFunction* obj,
                       std::function<double(const label_t*, int)> fn, data_size_t nd,
                       const data_size_t* bag, data_size_t bag_cnt,
                       const double* train_score) override {
    if (obj == nullptr || !obj->NeedRenewTreeOutput()) return;
    // every shard renews its local rows then syncs through the clique: all
    // copies end identical, take shard 0's outputs
    std::vector<std::unique_ptr<Tree>> copies(world_);
    for (int r = 0; r < world_; ++r) copies[r] = std::make_unique<Tree>(*tree);
    workers_->RunAll([&](int r) {
      learners_[r]->RenewTreeOutput(copies[r].get(), obj, fn, nd, bag, bag_cnt,
                                    train_score);
    });
    for (int l = 0; l < tree->num_leaves(); ++l)
      tree->SetLeafOutput(l, copies[0]->LeafOutput(l));
  }

  void DeviceBoosting(const ObjectiveFunction* obj) override {
    workers_->RunAll([&](int r) { learners_[r]->DeviceBoosting(obj); });
  }
  void DeviceAddInitScore(double v) override {
    workers_->RunAll([&](int r) { learners_[r]->DeviceAddInitScore(v); });
  }
  void SetClassOffset(int class_id) override {
    num_class_ = std::max(num_class_, class_id + 1);
    workers_->RunAll([&](int r) { learners_[r]->SetClassOffset(class_id); });
  }
  void DownloadTrainScore(double* dst) override {
    const int nc = std::max(1, config_->num_class);
    workers_->RunAll([&](int r) {
      const data_size_t sn = shard_begin_[r + 1] - shard_begin_[r];
      std::vector<double> tmp(static_cast<size_t>(sn) * nc);
      learners_[r]->DownloadTrainScore(tmp.data());
      for (int c = 0; c < nc; ++c)
        memcpy(dst + static_cast<size_t>(c) * num_data_ + shard_begin_[r],
               tmp.data() + static_cast<size_t>(c) * sn, sizeof(double) * sn);
    });
  }
  void UploadTrainScore(const double* src) override {
    const int nc = std::max(1, config_->num_class);
    workers_->RunAll([&](int r) {
      const data_size_t sn = shard_begin_[r + 1] - shard_begin_[r];
      std::vector<double> tmp(static_cast<size_t>(sn) * nc);
      for (int c = 0; c < nc; ++c)
        memcpy(tmp.data() + static_cast<size_t>(c) * sn,
               src + static_cast<size_t>(c) * num_data_ + shard_begin_[r],
               sizeof(double) * sn);
      learners_[r]->UploadTrainScore(tmp.data());
    });
  }
  bool DeviceEvalPointwise(int loss_kind, double loss_a, int convert_kind,
                           double convert_param, double* out_sum,
                           double* out_wsum) override {
    std::vector<double> sums(world_, 0.0), wsums(world_, 0.0);
    std::vector<int> oks(world_, 0);
    workers_->RunAll([&](int r) {
      oks[r] = learners_[r]->DeviceEvalPointwise(loss_kind, loss_a, convert_kind,
                                                 convert_param, &sums[r], &wsums[r])
                   ? 1
                   : 0;
    });
    for (int r = 0; r < world_; ++r)
      if (!oks[r]) return false;
    *out_sum = 0.0;
    *out_wsum = 0.0;
    for (int r = 0; r < world_; ++r) {
      *out_sum += sums[r];
      *out_wsum += wsums[r];
    }
    return true;
  }

 private:
  const Config* config_;
  const Dataset* train_data_ = nullptr;
  bool is_constant_hessian_ = false;
  data_size_t num_data_ = 0;
  int world_ = 1;
  int num_class_ = 1;
  std::vector<data_size_t> shard_begin_;
  std::vector<std::unique_ptr<Dataset>> shards_;
  std::vector<std::vector<data_size_t>> bag_parts_;  // per-shard local bag indices
  std::vector<std::unique_ptr<HIPTreeLearner>> learners_;
  std::unique_ptr<GpuWorkers> workers_;
  std::unique_ptr<InProcClique> clique_;
  std::vector<ncclComm_t> rccl_comms_;
};

// ------------------------------------------------------------------ registration
namespace {
TreeLearner* CreateHIP(const Config* cfg) {
  // loud, not silent: features the device split loop does not implement yet fall
  // back to the host serial learner (reference CUDA learner errors similarly)
  auto unsupported = [&]() -> const char* { return nullptr; };
  if (const char* what = unsupported()) {
    Log::Warning("device_type=%s: %s is not implemented in the HIP split loop; "
                 "training this model on the host (CPU) learner instead",
                 cfg->device_type.c_str(), what);
    return new SerialTreeLearner(cfg);
  }
  if (cfg->num_gpu > 1) {
    // single-process multi-GPU (reference NCCLGBDT parity): shard rows over
    // num_gpu device contexts inside this process, no torchrun needed
    return new MultiGpuTreeLearner(cfg);
  }
  return new HIPTreeLearner(cfg);
}
struct HIPRegistrar {
  HIPRegistrar() { g_create_hip_learner = CreateHIP; }
} hip_registrar;
}  // namespace

}  // namespace migbm

// ------------------------------------------------------------------ RCCL bootstrap C API
extern "C" __attribute__((visibility("default"))) int LGBM_GPUGetUniqueId(void* out_id,
                                                                          int* out_size) {
  try {
    ncclUniqueId id;
    NCCL_OK(ncclGetUniqueId(&id));
    memcpy(out_id, &id, sizeof(id));
    *out_size = sizeof(id);
    return 0;
  } catch (...) {
    return -1;
  }
}

extern "C" __attribute__((visibility("default"))) int LGBM_GPUNetworkInit(int world, int rank,
                                                                          const void* id_bytes) {
  try {
    using migbm::GpuComm;
    ncclUniqueId id;
    memcpy(&id, id_bytes, sizeof(id));
    auto& c = GpuComm::Get();
    NCCL_OK(ncclCommInitRank(&c.comm, world, id, rank));
    c.world = world;
    c.rank = rank;
    migbm::Log::Info("RCCL communicator initialized: rank %d / %d", rank, world);
    return 0;
  } catch (const std::exception& ex) {
    fprintf(stderr, "LGBM_GPUNetworkInit failed: %s\n", ex.what());
    return -1;
  } catch (...) {
    fprintf(stderr, "LGBM_GPUNetworkInit failed: unknown error\n");
    return -1;
  }
}

extern "C" __attribute__((visibility("default"))) int LGBM_GPUNetworkFree() {
  using migbm::GpuComm;
  auto& c = GpuComm::Get();
  if (c.comm) {
    ncclCommDestroy(c.comm);
    c.comm = nullptr;
  }
  c.world = 1;
  c.rank = 0;
  return 0;
}

extern "C" __attribute__((visibility("default"))) int LGBM_GPUSetDevice(int device) {
  return hipSetDevice(device) == hipSuccess ? 0 : -1;
}

/*! kernel-level unit-test hook (tests/test_gpu.py): device root histogram vs the
 *  fp64 host oracle, bin for bin. Not part of the public LGBM_* surface. */
extern "C" __attribute__((visibility("default"))) int MIGBM_DebugDeviceRootHist(
    void* dataset_handle, const char* params, const float* grad, const float* hess,
    double* out_max_rel_err) {
  try {
    auto* ds = static_cast<migbm::Dataset*>(dataset_handle);
    auto cfgmap = migbm::Config::Str2Map(params);
    migbm::Config cfg;
    cfg.Set(cfgmap);
    migbm::HIPTreeLearner learner(&cfg);
    learner.Init(ds, false);
    *out_max_rel_err = learner.DebugRootHistMaxRelErr(grad, hess);
    return 0;
  } catch (const std::exception& ex) {
    fprintf(stderr, "MIGBM_DebugDeviceRootHist: %s\n", ex.what());
    return -1;
  } catch (...) {
    return -1;
  }
}
