/*!
 * migbm HIP tree learner — full-GPU leaf-wise histogram learner for MI355X (gfx950).
 *
 * Fresh CDNA4-first design (capability target: the reference's CUDA single-GPU learner,
 * src/treelearner/cuda/* — see SURVEY.md §2.3; NOT a port):
 *  - binned feature matrix resident in HBM twice: row-major uint8 (histogram build,
 *    16B-aligned rows) and column-major (partition decisions, coalesced)
 *  - ConstructHistogram: LDS-privatized per-workgroup histograms with DS atomics,
 *    feature-partitioned when the histogram exceeds the LDS budget, grid-stride rows
 *  - histogram subtraction for the larger child (parent -= smaller, in place)
 *  - best-split: one wave64 per feature, shfl-based inclusive prefix scan over bins,
 *    both missing-direction variants, fp64 gain math identical to the CPU oracle
 *  - data partition: mark -> block scan -> wave-ballot ranked scatter (stable)
 *  - per-split host sync: 2 small D2H reads (winner record; exact left count)
 *  - device-resident boosting: objective gradient kernels (binary logloss, L2) run on
 *    the device score vector; other objectives fall back to host gradients transparently
 *  - multi-GPU data parallelism: RCCL allreduce of leaf histograms/root sums over xGMI
 *    (one process per GPU; comm injected via LGBM_GPUNetworkInit)
 */
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "migbm/tree_learner.h"
#include "migbm/objective.h"

#include <algorithm>

namespace migbm {

#define HIP_OK(call)                                                                  \
  do {                                                                                \
    hipError_t _e = (call);                                                           \
    if (_e != hipSuccess)                                                             \
      Log::Fatal("HIP error %s at %s:%d", hipGetErrorString(_e), __FILE__, __LINE__); \
  } while (0)

#define NCCL_OK(call)                                                                  \
  do {                                                                                 \
    ncclResult_t _e = (call);                                                          \
    if (_e != ncclSuccess)                                                             \
      ::migbm::Log::Fatal("RCCL error %s at %s:%d", ncclGetErrorString(_e), __FILE__,  \
                          __LINE__);                                                   \
  } while (0)

// ------------------------------------------------------------------ GPU comm singleton
struct GpuComm {
  ncclComm_t comm = nullptr;
  int world = 1;
  int rank = 0;
  bool active() const { return comm != nullptr && world > 1; }
  static GpuComm& Get() {
    static GpuComm c;
    return c;
  }
};

// ------------------------------------------------------------------ device structs
namespace hipk {

struct SplitRec {
  double gain;
  double left_g, left_h;
  double left_out, right_out;
  int left_cnt, right_cnt;  // hessian-derived (approx)
  int feature;
  int bin;
  int default_left;
  int valid;
};

struct LeafStat {
  double sum_g, sum_h;
  int cnt;
  int pad;
};

struct GainParams {
  double l1, l2, mds;
  double min_hess, min_gain_shift_add;  // min_gain_to_split
  int min_data;
  int max_cat_to_onehot;
};

struct FeatMeta {
  int bin_off;          // global histogram bin offset
  int num_bin;
  int num_numeric_bin;  // bins eligible as thresholds
  int nan_bin;          // -1 if none
  int is_cat;
};

__device__ __forceinline__ double d_thl1(double s, double l1) {
  double r = fabs(s) - l1;
  r = r > 0.0 ? r : 0.0;
  return s >= 0.0 ? r : -r;
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
__device__ __forceinline__ double d_leaf_gain(double g, double h, const GainParams& p) {
  if (p.mds <= 0.0) {
    double s = d_thl1(g, p.l1);
    return s * s / (h + p.l2);
  }
  double out = d_leaf_out(g, h, p);
  return d_gain_out(g, h, out, p);
}

__global__ void k_iota(uint32_t* p, int n);

// ------------------------------------------------------------------ histogram
/*! LDS-privatized histogram over one feature range [feat_begin, feat_end).
 *  part_bin_base/part_bins select the slice of the leaf histogram covered here. */
__global__ void k_hist(const uint8_t* __restrict__ rows, int stride,
                       const uint32_t* __restrict__ idx, int cnt,
                       const float* __restrict__ g, const float* __restrict__ h,
                       const FeatMeta* __restrict__ fm, int feat_begin, int feat_end,
                       int part_bin_base, int part_bins, float* __restrict__ ghist) {
  extern __shared__ float lh[];
  const int nelem = part_bins * 2;
  for (int i = threadIdx.x; i < nelem; i += blockDim.x) lh[i] = 0.0f;
  __syncthreads();
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  const int nthreads = blockDim.x * gridDim.x;
  for (int i = tid; i < cnt; i += nthreads) {
    const uint32_t r = idx[i];
    const float gi = g[r];
    const float hi = h[r];
    const uint8_t* rp = rows + static_cast<size_t>(r) * stride;
    for (int f = feat_begin; f < feat_end; ++f) {
      const int b = rp[f];
      const int off = (fm[f].bin_off - part_bin_base + b) * 2;
      atomicAdd(&lh[off], gi);
      atomicAdd(&lh[off + 1], hi);
    }
  }
  __syncthreads();
  float* gh = ghist + static_cast<size_t>(part_bin_base) * 2;
  for (int i = threadIdx.x; i < nelem; i += blockDim.x) {
    const float v = lh[i];
    if (v != 0.0f) atomicAdd(&gh[i], v);
  }
}

__global__ void k_hist_zero(float* hist, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) hist[i] = 0.0f;
}

__global__ void k_hist_subtract(float* __restrict__ parent, const float* __restrict__ small,
                                int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) parent[i] -= small[i];
}

// ------------------------------------------------------------------ reductions
__global__ void k_root_sums(const uint32_t* __restrict__ idx, int cnt,
                            const float* __restrict__ g, const float* __restrict__ h,
                            LeafStat* stat, int exact_cnt) {
  __shared__ double sg[8], sh[8];
  double tg = 0, th = 0;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = tid; i < cnt; i += blockDim.x * gridDim.x) {
    const uint32_t r = idx[i];
    tg += g[r];
    th += h[r];
  }
  // wave reduce
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
    if (blockIdx.x == 0) stat->cnt = exact_cnt;
  }
}

__global__ void k_zero_stat(LeafStat* stat) {
  stat->sum_g = 0;
  stat->sum_h = 0;
  stat->cnt = 0;
}

// ------------------------------------------------------------------ best split
/*! one wave (64 threads) per feature; scans numeric thresholds in both missing
 *  directions; one-hot scan for categorical features. */
__global__ void __launch_bounds__(64) k_best_feat(
    const float* __restrict__ hist, const FeatMeta* __restrict__ fm, int nf,
    const LeafStat* __restrict__ stat, GainParams p, const int8_t* __restrict__ feat_mask,
    SplitRec* __restrict__ out) {
  const int f = blockIdx.x;
  if (f >= nf) return;
  const int lane = threadIdx.x;
  SplitRec& rec = out[f];
  const FeatMeta m = fm[f];
  if (lane == 0) {
    rec.valid = 0;
    rec.feature = f;
  }
  if (feat_mask != nullptr && !feat_mask[f]) return;
  const double sum_g = stat->sum_g;
  const double sum_h = stat->sum_h;
  const int num_data = stat->cnt;
  if (num_data < 2 * p.min_data) return;
  const double cnt_factor = (num_data > 0 && sum_h > 0) ? num_data / sum_h : 1.0;
  const double parent_gain = d_leaf_gain(sum_g, sum_h, p);
  const double min_gain_shift = parent_gain + p.min_gain_shift_add;

  const float* fh = hist + static_cast<size_t>(m.bin_off) * 2;
  double g_nan = 0, h_nan = 0;
  const bool has_nan = m.nan_bin >= 0;
  if (has_nan) {
    g_nan = fh[2 * m.nan_bin];
    h_nan = fh[2 * m.nan_bin + 1];
  }

  double best_gain = -1e308;
  int best_bin = -1, best_dl = 0;
  double best_lg = 0, best_lh = 0;

  if (m.is_cat) {
    // one-hot scan: left = single bin b
    for (int b = lane; b < m.num_bin; b += 64) {
      const double gl = fh[2 * b], hl = fh[2 * b + 1];
      const double gr = sum_g - gl, hr = sum_h - hl;
      const int lc = static_cast<int>(hl * cnt_factor + 0.5);
      const int rc = num_data - lc;
      if (hl < p.min_hess || lc < p.min_data || hr < p.min_hess || rc < p.min_data) continue;
      const double lo = d_leaf_out(gl, hl, p), ro = d_leaf_out(gr, hr, p);
      const double gain = d_gain_out(gl, hl, lo, p) + d_gain_out(gr, hr, ro, p);
      if (gain <= min_gain_shift) continue;
      if (gain > best_gain) {
        best_gain = gain;
        best_bin = b;
        best_dl = 0;
        best_lg = gl;
        best_lh = hl;
      }
    }
  } else {
    const int nb = m.num_numeric_bin;
    const int t_max = nb - 2;
    double carry_g = 0, carry_h = 0;
    for (int chunk = 0; chunk * 64 < nb; ++chunk) {
      const int b = chunk * 64 + lane;
      double gb = b < nb ? static_cast<double>(fh[2 * b]) : 0.0;
      double hb = b < nb ? static_cast<double>(fh[2 * b + 1]) : 0.0;
      // inclusive wave scan
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
        const double lo = d_leaf_out(gl, hl, p), ro = d_leaf_out(gr, hr, p);
        const double gain = d_gain_out(gl, hl, lo, p) + d_gain_out(gr, hr, ro, p);
        if (gain <= min_gain_shift) continue;
        // prefer smaller bin on exact ties (matches CPU scan order)
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
  // wave argmax (gain desc, bin asc on tie)
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
    rec.gain = best_gain - min_gain_shift + p.min_gain_shift_add;  // improvement over parent
    rec.feature = f;
    rec.bin = best_bin;
    rec.default_left = best_dl;
    rec.left_g = best_lg;
    rec.left_h = best_lh;
    const double rg = sum_g - best_lg, rh = sum_h - best_lh;
    rec.left_out = d_leaf_out(best_lg, best_lh, p);
    rec.right_out = d_leaf_out(rg, rh, p);
    rec.left_cnt = static_cast<int>(best_lh * cnt_factor + 0.5);
    rec.right_cnt = num_data - rec.left_cnt;
  }
}

/*! reduce per-feature records to one per leaf (gain desc, feature asc on tie). */
__global__ void k_best_leaf(const SplitRec* __restrict__ feat_best, int nf,
                            SplitRec* __restrict__ leaf_best, const LeafStat* stat) {
  // single block
  __shared__ int s_idx[256];
  __shared__ double s_gain[256];
  const int tid = threadIdx.x;
  int bi = -1;
  double bg = -1e308;
  for (int f = tid; f < nf; f += blockDim.x) {
    if (feat_best[f].valid && (feat_best[f].gain > bg ||
                               (feat_best[f].gain == bg && (bi < 0 || f < bi)))) {
      bg = feat_best[f].gain;
      bi = f;
    }
  }
  s_idx[tid] = bi;
  s_gain[tid] = bg;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (tid < s) {
      if (s_idx[tid + s] >= 0 &&
          (s_idx[tid] < 0 || s_gain[tid + s] > s_gain[tid] ||
           (s_gain[tid + s] == s_gain[tid] && s_idx[tid + s] < s_idx[tid]))) {
        s_idx[tid] = s_idx[tid + s];
        s_gain[tid] = s_gain[tid + s];
      }
    }
    __syncthreads();
  }
  if (tid == 0) {
    if (s_idx[0] >= 0) {
      *leaf_best = feat_best[s_idx[0]];
      // stash parent totals in unused fields for host-side child stats
      leaf_best->right_cnt = stat->cnt - leaf_best->left_cnt;
    } else {
      leaf_best->valid = 0;
      leaf_best->gain = -1e308;
    }
  }
}

/*! argmax over current leaves' best records -> winner (leaf idx in rec.right_cnt slot
 *  abused? no: separate output int). */
__global__ void k_best_overall(const SplitRec* __restrict__ leaf_best, int num_leaves,
                               SplitRec* __restrict__ winner, int* __restrict__ winner_leaf) {
  __shared__ int s_idx[256];
  __shared__ double s_gain[256];
  const int tid = threadIdx.x;
  int bi = -1;
  double bg = 0.0;  // require gain > 0
  for (int l = tid; l < num_leaves; l += blockDim.x) {
    if (leaf_best[l].valid && leaf_best[l].gain > bg) {
      bg = leaf_best[l].gain;
      bi = l;
    }
  }
  s_idx[tid] = bi;
  s_gain[tid] = bg;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (tid < s) {
      if (s_idx[tid + s] >= 0 &&
          (s_idx[tid] < 0 || s_gain[tid + s] > s_gain[tid] ||
           (s_gain[tid + s] == s_gain[tid] && s_idx[tid + s] < s_idx[tid]))) {
        s_idx[tid] = s_idx[tid + s];
        s_gain[tid] = s_gain[tid + s];
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
/*! decision mark per row of the leaf (1 = left). */
__global__ void k_part_mark(const uint8_t* __restrict__ colbins, const uint32_t* __restrict__ idx,
                            int cnt, int thr_bin, int nan_bin, int default_left,
                            int cat_onehot, uint8_t* __restrict__ marks,
                            int* __restrict__ block_cnt) {
  __shared__ int s_cnt[8];
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  int go = 0;
  if (i < cnt) {
    const int b = colbins[idx[i]];
    if (cat_onehot) go = b == thr_bin ? 1 : 0;
    else if (nan_bin >= 0 && b == nan_bin) go = default_left;
    else go = b <= thr_bin ? 1 : 0;
    marks[i] = static_cast<uint8_t>(go);
  }
  // wave count
  const uint64_t ball = __ballot(go);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) s_cnt[wave] = __popcll(ball);
  __syncthreads();
  if (threadIdx.x == 0) {
    int c = 0;
    for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) c += s_cnt[w];
    block_cnt[blockIdx.x] = c;
  }
}

/*! categorical variant: left iff bin in bitset. */
__global__ void k_part_mark_cat(const uint8_t* __restrict__ colbins,
                                const uint32_t* __restrict__ idx, int cnt,
                                const uint32_t* __restrict__ bits, int n_words,
                                uint8_t* __restrict__ marks, int* __restrict__ block_cnt) {
  __shared__ int s_cnt[8];
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  int go = 0;
  if (i < cnt) {
    const int b = colbins[idx[i]];
    go = (b >> 5) < n_words && ((bits[b >> 5] >> (b & 31)) & 1) ? 1 : 0;
    marks[i] = static_cast<uint8_t>(go);
  }
  const uint64_t ball = __ballot(go);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) s_cnt[wave] = __popcll(ball);
  __syncthreads();
  if (threadIdx.x == 0) {
    int c = 0;
    for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) c += s_cnt[w];
    block_cnt[blockIdx.x] = c;
  }
}

/*! exclusive scan of block left counts; emits per-block left offset and total. */
__global__ void k_part_scan(const int* __restrict__ block_cnt, int nblocks,
                            int* __restrict__ block_off, int* __restrict__ total_left,
                            int cnt) {
  // single block, serial-ish scan with 256-wide chunks
  __shared__ int carry;
  if (threadIdx.x == 0) carry = 0;
  __syncthreads();
  for (int base = 0; base < nblocks; base += blockDim.x) {
    const int i = base + threadIdx.x;
    int v = i < nblocks ? block_cnt[i] : 0;
    // inclusive scan within wave then across waves via shared
    __shared__ int s_w[8];
    int incl = v;
    for (int d = 1; d < 64; d <<= 1) {
      const int t = __shfl_up(incl, d);
      if ((threadIdx.x & 63) >= static_cast<unsigned>(d)) incl += t;
    }
    const int wave = threadIdx.x / 64;
    if ((threadIdx.x & 63) == 63) s_w[wave] = incl;
    __syncthreads();
    int wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += s_w[w];
    const int excl = carry + wave_base + incl - v;
    if (i < nblocks) block_off[i] = excl;
    __syncthreads();
    if (threadIdx.x == blockDim.x - 1) carry = carry + wave_base + incl;
    __syncthreads();
  }
  if (threadIdx.x == 0) *total_left = carry;
  (void)cnt;
}

/*! stable scatter into tmp using marks + block offsets. */
__global__ void k_part_scatter(const uint32_t* __restrict__ idx, int cnt,
                               const uint8_t* __restrict__ marks,
                               const int* __restrict__ block_off,
                               const int* __restrict__ total_left,
                               uint32_t* __restrict__ out) {
  __shared__ int s_left[8], s_tot[8];
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int go = i < cnt ? marks[i] : 0;
  const uint64_t ball = __ballot(go);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x / 64;
  const int lane_rank_left = __popcll(ball & ((1ull << lane) - 1));
  if (lane == 63) s_left[wave] = __popcll(ball);
  __syncthreads();
  int wave_left_base = 0;
  for (int w = 0; w < wave; ++w) wave_left_base += s_left[w];
  int block_left_total = 0;
  for (int w = 0; w < static_cast<int>(blockDim.x / 64); ++w) block_left_total += s_left[w];
  (void)s_tot;
  if (i >= cnt) return;
  const int tl = *total_left;
  const int bl_off = block_off[blockIdx.x];
  const int valid_in_block = min(static_cast<int>(blockDim.x), cnt - blockIdx.x * static_cast<int>(blockDim.x));
  if (go) {
    const int pos = bl_off + wave_left_base + lane_rank_left;
    out[pos] = idx[i];
  } else {
    // rights: block's right offset = (block start - lefts before block)
    const int rights_before_block = blockIdx.x * blockDim.x - bl_off;
    const int lane_rank_right = lane - lane_rank_left;
    int wave_right_base = wave * 64 - wave_left_base;
    // clamp for the tail wave
    (void)valid_in_block;
    const int pos = tl + rights_before_block + wave_right_base + lane_rank_right;
    out[pos] = idx[i];
  }
}

__global__ void k_copy_idx(const uint32_t* __restrict__ src, uint32_t* __restrict__ dst,
                           int cnt) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < cnt) dst[i] = src[i];
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

__global__ void k_score_add_const(double* score, int n, double v) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) score[i] += v;
}

/*! score[row] += leaf_output[leaf-of-position] using the final partition layout. */
__global__ void k_score_update(const uint32_t* __restrict__ idx,
                               const int* __restrict__ leaf_begin,
                               const int* __restrict__ leaf_cnt, int num_leaves, int used_cnt,
                               const double* __restrict__ leaf_out, double* __restrict__ score) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= used_cnt) return;
  // binary search leaf via begin array (leaves are ordered segments)
  int lo = 0, hi = num_leaves - 1;
  while (lo < hi) {
    const int mid = (lo + hi + 1) >> 1;
    if (i >= leaf_begin[mid]) lo = mid;
    else hi = mid - 1;
  }
  // account for segment gaps (leaf segments are contiguous by construction)
  score[idx[i]] += leaf_out[lo];
  (void)leaf_cnt;
}

/*! device tree walk over column bins (OOB score update). */
__global__ void k_tree_predict_add(const uint8_t* const* __restrict__ cols,
                                   const int* __restrict__ split_feat,
                                   const int* __restrict__ thr_bin,
                                   const int* __restrict__ left_child,
                                   const int* __restrict__ right_child,
                                   const int* __restrict__ nan_bin,
                                   const uint8_t* __restrict__ default_left,
                                   const double* __restrict__ leaf_out,
                                   const uint32_t* __restrict__ rows, int n,
                                   double* __restrict__ score) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const uint32_t r = rows[i];
  int node = 0;
  while (node >= 0) {
    const int f = split_feat[node];
    const int b = cols[f][r];
    if (nan_bin[node] >= 0 && b == nan_bin[node]) {
      node = default_left[node] ? left_child[node] : right_child[node];
    } else {
      node = b <= thr_bin[node] ? left_child[node] : right_child[node];
    }
  }
  score[r] += leaf_out[~node];
}

}  // namespace hipk

// ------------------------------------------------------------------ RAII device buffer
template <typename T>
struct DevBuf {
  T* ptr = nullptr;
  size_t n = 0;
  void Alloc(size_t count) {
    if (count == n) return;
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
  operator T*() { return ptr; }
};

// ------------------------------------------------------------------ the learner
class HIPTreeLearner : public TreeLearner {
 public:
  explicit HIPTreeLearner(const Config* config) : config_(config) {}
  ~HIPTreeLearner() override {
    if (stream_) hipStreamDestroy(stream_);
  }

  bool IsHIPLearner() const override { return true; }
  bool DeviceObjectiveSupported(const std::string& name) const override {
    return name == "binary" || name == "regression";
  }

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
    const int n = num_data_;
    hipLaunchKernelGGL(hipk::k_score_add_const, dim3((n + 255) / 256), dim3(256), 0, stream_,
                       d_score_, n, v);
  }
  void DownloadTrainScore(double* dst) override {
    HIP_OK(hipStreamSynchronize(stream_));
    HIP_OK(hipMemcpy(dst, d_score_.ptr, sizeof(double) * num_data_, hipMemcpyDeviceToHost));
  }

 private:
  void UploadGradients(const score_t* g, const score_t* h);
  void BuildHistogram(int slot, int begin, int cnt);
  void ReduceHistogram(int slot);  // RCCL allreduce over xGMI (multi-GPU)
  void LaunchBestSplit(int leaf, int slot);
  int PartitionLeaf(int leaf, const hipk::SplitRec& rec, const SplitInfo* cat_split);

  const Config* config_;
  const Dataset* train_data_ = nullptr;
  bool is_constant_hessian_ = false;
  hipStream_t stream_ = nullptr;

  int num_data_ = 0;
  int nf_ = 0;
  int total_bins_ = 0;
  int row_stride_ = 0;
  std::vector<hipk::FeatMeta> feat_meta_host_;
  std::vector<std::pair<int, int>> feat_partitions_;  // [begin,end) feature ranges for LDS
  std::vector<std::pair<int, int>> part_bin_range_;   // bin [base,count) per partition

  DevBuf<uint8_t> d_rows_;        // row-major bins
  DevBuf<uint8_t> d_cols_;        // column-major bins (nf * num_data)
  DevBuf<uint8_t*> d_col_ptrs_;
  DevBuf<hipk::FeatMeta> d_feat_meta_;
  DevBuf<float> d_grad_, d_hess_;
  DevBuf<double> d_score_;
  DevBuf<float> d_label_, d_weight_;
  DevBuf<uint32_t> d_idx_, d_idx_tmp_;
  DevBuf<uint8_t> d_marks_;
  DevBuf<int> d_block_cnt_, d_block_off_, d_total_left_;
  DevBuf<float> d_hist_;          // num_leaves slots
  DevBuf<hipk::SplitRec> d_feat_best_;
  DevBuf<hipk::SplitRec> d_leaf_best_;
  DevBuf<hipk::SplitRec> d_winner_;
  DevBuf<int> d_winner_leaf_;
  DevBuf<hipk::LeafStat> d_leaf_stats_;
  DevBuf<int8_t> d_feat_mask_;
  DevBuf<int> d_leaf_begin_, d_leaf_cnt_;
  DevBuf<double> d_leaf_out_;
  // tree-walk buffers (OOB updates)
  DevBuf<int> d_tw_feat_, d_tw_thr_, d_tw_left_, d_tw_right_, d_tw_nan_;
  DevBuf<uint8_t> d_tw_dl_;
  DevBuf<double> d_tw_out_;
  DevBuf<uint32_t> d_oob_;

  bool grads_on_device_ = false;
  bool weights_present_ = false;
  const data_size_t* bag_indices_ = nullptr;
  data_size_t bag_cnt_ = 0;
  data_size_t used_cnt_ = 0;

  // host mirrors
  std::vector<int> leaf_begin_, leaf_cnt_, leaf_slot_;
  std::vector<SplitInfo> cpu_cat_split_;  // per-leaf categorical fallback info
  Random feature_rng_{0};
  std::vector<int8_t> feat_mask_host_;

  static constexpr int kHistBlock = 256;
  static constexpr int kLdsBudget = 64 * 1024;  // bytes of LDS per hist block
};

void HIPTreeLearner::Init(const Dataset* train_data, bool is_constant_hessian) {
  train_data_ = train_data;
  is_constant_hessian_ = is_constant_hessian;
  num_data_ = train_data->num_data();
  nf_ = train_data->num_features();
  total_bins_ = train_data->num_total_bin();
  feature_rng_ = Random(config_->feature_fraction_seed);

  if (!stream_) HIP_OK(hipStreamCreate(&stream_));

  // feature metadata
  feat_meta_host_.resize(nf_);
  for (int f = 0; f < nf_; ++f) {
    const BinMapper* m = train_data->FeatureBinMapper(f);
    if (m->num_bin() > 256) Log::Fatal("HIP learner currently supports max_bin<=255");
    feat_meta_host_[f] = {static_cast<int>(train_data->hist_offset(f)), m->num_bin(),
                          m->num_numeric_bin(), m->nan_bin(),
                          m->bin_type() == BinType::kCategorical ? 1 : 0};
  }
  d_feat_meta_.Alloc(nf_);
  HIP_OK(hipMemcpy(d_feat_meta_.ptr, feat_meta_host_.data(), sizeof(hipk::FeatMeta) * nf_,
                   hipMemcpyHostToDevice));

  // LDS feature partitioning: contiguous feature ranges whose bins fit the LDS budget
  feat_partitions_.clear();
  part_bin_range_.clear();
  const int max_bins_per_part = kLdsBudget / (2 * sizeof(float));
  int begin = 0;
  while (begin < nf_) {
    int end = begin;
    int bins = 0;
    while (end < nf_ && bins + feat_meta_host_[end].num_bin <= max_bins_per_part) {
      bins += feat_meta_host_[end].num_bin;
      ++end;
    }
    if (end == begin) Log::Fatal("Single feature exceeds LDS histogram budget");
    feat_partitions_.emplace_back(begin, end);
    part_bin_range_.emplace_back(feat_meta_host_[begin].bin_off, bins);
    begin = end;
  }

  // data upload: row-major + column-major
  const auto& view = train_data->GetRowMajorView();
  if (view.is16) Log::Fatal("HIP learner currently supports max_bin<=255 (uint8 bins)");
  row_stride_ = view.row_stride;
  d_rows_.Alloc(view.data.size());
  HIP_OK(hipMemcpy(d_rows_.ptr, view.data.data(), view.data.size(), hipMemcpyHostToDevice));
  d_cols_.Alloc(static_cast<size_t>(nf_) * num_data_);
  std::vector<uint8_t*> col_ptrs(nf_);
  for (int f = 0; f < nf_; ++f) {
    uint8_t* dst = d_cols_.ptr + static_cast<size_t>(f) * num_data_;
    HIP_OK(hipMemcpy(dst, train_data->column(f).data8(), num_data_, hipMemcpyHostToDevice));
    col_ptrs[f] = dst;
  }
  d_col_ptrs_.Alloc(nf_);
  HIP_OK(hipMemcpy(d_col_ptrs_.ptr, col_ptrs.data(), sizeof(uint8_t*) * nf_,
                   hipMemcpyHostToDevice));

  // working buffers
  d_grad_.Alloc(num_data_);
  d_hess_.Alloc(num_data_);
  d_score_.Alloc(num_data_);
  HIP_OK(hipMemset(d_score_.ptr, 0, sizeof(double) * num_data_));
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
  const int max_blocks = (num_data_ + kHistBlock - 1) / kHistBlock + 1;
  d_block_cnt_.Alloc(max_blocks);
  d_block_off_.Alloc(max_blocks);
  d_total_left_.Alloc(1);
  const int nl = config_->num_leaves;
  d_hist_.Alloc(static_cast<size_t>(nl) * total_bins_ * 2);
  d_feat_best_.Alloc(static_cast<size_t>(nf_));
  d_leaf_best_.Alloc(nl);
  d_winner_.Alloc(1);
  d_winner_leaf_.Alloc(1);
  d_leaf_stats_.Alloc(nl);
  d_feat_mask_.Alloc(nf_);
  d_leaf_begin_.Alloc(nl);
  d_leaf_cnt_.Alloc(nl);
  d_leaf_out_.Alloc(nl);
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
  leaf_slot_.resize(nl);
  Log::Info("HIP tree learner initialized: %d rows, %d features, %d bins, %zu LDS partitions",
            num_data_, nf_, total_bins_, feat_partitions_.size());
}

void HIPTreeLearner::UploadGradients(const score_t* g, const score_t* h) {
  HIP_OK(hipMemcpyAsync(d_grad_.ptr, g, sizeof(float) * num_data_, hipMemcpyHostToDevice,
                        stream_));
  HIP_OK(hipMemcpyAsync(d_hess_.ptr, h, sizeof(float) * num_data_, hipMemcpyHostToDevice,
                        stream_));
}

void HIPTreeLearner::DeviceBoosting(const ObjectiveFunction* obj) {
  const std::string name = obj->GetName();
  const int n = num_data_;
  const dim3 g((n + 255) / 256), b(256);
  if (name == "binary") {
    // mirror BinaryLogloss's label weights
    double w_pos = config_->scale_pos_weight, w_neg = 1.0;
    if (config_->is_unbalance) {
      data_size_t pos = obj->NumPositiveData();
      data_size_t neg = num_data_ - pos;
      if (pos > 0 && neg > 0) {
        w_pos = pos > neg ? 1.0 : static_cast<double>(neg) / pos;
        w_neg = pos > neg ? static_cast<double>(pos) / neg : 1.0;
      }
    }
    hipLaunchKernelGGL(hipk::k_grad_binary, g, b, 0, stream_, d_score_, d_label_,
                       weights_present_ ? d_weight_.ptr : nullptr, n, config_->sigmoid,
                       w_pos, w_neg, d_grad_, d_hess_);
  } else if (name == "regression") {
    hipLaunchKernelGGL(hipk::k_grad_l2, g, b, 0, stream_, d_score_, d_label_,
                       weights_present_ ? d_weight_.ptr : nullptr, n, d_grad_, d_hess_);
  } else {
    Log::Fatal("DeviceBoosting called for unsupported objective %s", name.c_str());
  }
  grads_on_device_ = true;
}

void HIPTreeLearner::BuildHistogram(int slot, int begin, int cnt) {
  float* hist = d_hist_.ptr + static_cast<size_t>(slot) * total_bins_ * 2;
  const int n_elem = total_bins_ * 2;
  hipLaunchKernelGGL(hipk::k_hist_zero, dim3((n_elem + 255) / 256), dim3(256), 0, stream_,
                     hist, n_elem);
  // grid: enough blocks to fill 256 CUs x 8 XCDs; grid-stride handles the rest
  int blocks = std::min(4096, std::max(1, (cnt + kHistBlock * 8 - 1) / (kHistBlock * 8)));
  for (size_t p = 0; p < feat_partitions_.size(); ++p) {
    const auto [fb, fe] = feat_partitions_[p];
    const auto [bin_base, bins] = part_bin_range_[p];
    hipLaunchKernelGGL(hipk::k_hist, dim3(blocks), dim3(kHistBlock),
                       bins * 2 * sizeof(float), stream_, d_rows_.ptr, row_stride_,
                       d_idx_.ptr + begin, cnt, d_grad_, d_hess_, d_feat_meta_, fb, fe,
                       bin_base, bins, hist);
  }
}

void HIPTreeLearner::ReduceHistogram(int slot) {
  auto& comm = GpuComm::Get();
  if (!comm.active()) return;
  float* hist = d_hist_.ptr + static_cast<size_t>(slot) * total_bins_ * 2;
  NCCL_OK(ncclAllReduce(hist, hist, static_cast<size_t>(total_bins_) * 2, ncclFloat32,
                        ncclSum, comm.comm, stream_));
}

void HIPTreeLearner::LaunchBestSplit(int leaf, int slot) {
  hipk::GainParams p;
  p.l1 = config_->lambda_l1;
  p.l2 = config_->lambda_l2;
  p.mds = config_->max_delta_step;
  p.min_hess = config_->min_sum_hessian_in_leaf;
  p.min_gain_shift_add = config_->min_gain_to_split;
  p.min_data = config_->min_data_in_leaf;
  p.max_cat_to_onehot = config_->max_cat_to_onehot;
  float* hist = d_hist_.ptr + static_cast<size_t>(slot) * total_bins_ * 2;
  hipLaunchKernelGGL(hipk::k_best_feat, dim3(nf_), dim3(64), 0, stream_, hist, d_feat_meta_,
                     nf_, d_leaf_stats_.ptr + leaf, p,
                     feat_mask_host_.empty() ? nullptr : d_feat_mask_.ptr, d_feat_best_.ptr);
  hipLaunchKernelGGL(hipk::k_best_leaf, dim3(1), dim3(256), 0, stream_, d_feat_best_.ptr, nf_,
                     d_leaf_best_.ptr + leaf, d_leaf_stats_.ptr + leaf);
}

int HIPTreeLearner::PartitionLeaf(int leaf, const hipk::SplitRec& rec, const SplitInfo*) {
  const int begin = leaf_begin_[leaf];
  const int cnt = leaf_cnt_[leaf];
  const int nblocks = (cnt + kHistBlock - 1) / kHistBlock;
  const hipk::FeatMeta& m = feat_meta_host_[rec.feature];
  const uint8_t* col = d_cols_.ptr + static_cast<size_t>(rec.feature) * num_data_;
  hipLaunchKernelGGL(hipk::k_part_mark, dim3(nblocks), dim3(kHistBlock), 0, stream_, col,
                     d_idx_.ptr + begin, cnt, rec.bin, m.is_cat ? -1 : m.nan_bin,
                     rec.default_left, m.is_cat ? 1 : 0, d_marks_.ptr, d_block_cnt_.ptr);
  hipLaunchKernelGGL(hipk::k_part_scan, dim3(1), dim3(256), 0, stream_, d_block_cnt_.ptr,
                     nblocks, d_block_off_.ptr, d_total_left_.ptr, cnt);
  hipLaunchKernelGGL(hipk::k_part_scatter, dim3(nblocks), dim3(kHistBlock), 0, stream_,
                     d_idx_.ptr + begin, cnt, d_marks_.ptr, d_block_off_.ptr,
                     d_total_left_.ptr, d_idx_tmp_.ptr + begin);
  hipLaunchKernelGGL(hipk::k_copy_idx, dim3(nblocks), dim3(kHistBlock), 0, stream_,
                     d_idx_tmp_.ptr + begin, d_idx_.ptr + begin, cnt);
  int total_left = 0;
  HIP_OK(hipMemcpyAsync(&total_left, d_total_left_.ptr, sizeof(int), hipMemcpyDeviceToHost,
                        stream_));
  HIP_OK(hipStreamSynchronize(stream_));
  return total_left;
}

Tree* HIPTreeLearner::Train(const score_t* gradients, const score_t* hessians, bool) {
  const int nl = config_->num_leaves;
  auto tree = std::make_unique<Tree>(nl);

  if (!grads_on_device_) UploadGradients(gradients, hessians);
  grads_on_device_ = false;

  // feature sampling mask (per tree)
  feat_mask_host_.clear();
  if (config_->feature_fraction < 1.0) {
    feat_mask_host_.assign(nf_, 0);
    int k = std::max(1, static_cast<int>(nf_ * config_->feature_fraction));
    for (int f : feature_rng_.Sample(nf_, k)) feat_mask_host_[f] = 1;
    HIP_OK(hipMemcpyAsync(d_feat_mask_.ptr, feat_mask_host_.data(), nf_,
                          hipMemcpyHostToDevice, stream_));
  }

  // root: indices = bag or iota
  if (bag_indices_ != nullptr && bag_cnt_ > 0) {
    used_cnt_ = bag_cnt_;
    HIP_OK(hipMemcpyAsync(d_idx_.ptr, bag_indices_, sizeof(uint32_t) * bag_cnt_,
                          hipMemcpyHostToDevice, stream_));
  } else {
    used_cnt_ = num_data_;
    hipLaunchKernelGGL(hipk::k_iota, dim3((num_data_ + 255) / 256), dim3(256), 0, stream_,
                       d_idx_.ptr, num_data_);
  }
  std::fill(leaf_begin_.begin(), leaf_begin_.end(), 0);
  std::fill(leaf_cnt_.begin(), leaf_cnt_.end(), 0);
  leaf_begin_[0] = 0;
  leaf_cnt_[0] = used_cnt_;
  for (int l = 0; l < nl; ++l) leaf_slot_[l] = l;

  // root stats
  hipLaunchKernelGGL(hipk::k_zero_stat, dim3(1), dim3(1), 0, stream_, d_leaf_stats_.ptr);
  {
    int blocks = std::min(2048, (static_cast<int>(used_cnt_) + 255) / 256);
    hipLaunchKernelGGL(hipk::k_root_sums, dim3(blocks), dim3(256), 0, stream_, d_idx_.ptr,
                       used_cnt_, d_grad_, d_hess_, d_leaf_stats_.ptr,
                       static_cast<int>(used_cnt_));
  }
  // multi-GPU: globalize root stats (sum over ranks; exact global count)
  auto& comm = GpuComm::Get();
  if (comm.active()) {
    NCCL_OK(ncclAllReduce(d_leaf_stats_.ptr, d_leaf_stats_.ptr, 2, ncclFloat64, ncclSum,
                          comm.comm, stream_));
    // count is int in the struct; reduce separately via host (cheap, once per tree)
    int64_t cnt = used_cnt_;
    // piggyback on RCCL with a small device buffer
    static int64_t* d_cnt = nullptr;
    if (!d_cnt) HIP_OK(hipMalloc(&d_cnt, sizeof(int64_t)));
    HIP_OK(hipMemcpyAsync(d_cnt, &cnt, sizeof(int64_t), hipMemcpyHostToDevice, stream_));
    NCCL_OK(ncclAllReduce(d_cnt, d_cnt, 1, ncclInt64, ncclSum, comm.comm, stream_));
    int64_t global_cnt = 0;
    HIP_OK(hipMemcpyAsync(&global_cnt, d_cnt, sizeof(int64_t), hipMemcpyDeviceToHost,
                          stream_));
    HIP_OK(hipStreamSynchronize(stream_));
    // overwrite stat count with the global one
    hipk::LeafStat st;
    HIP_OK(hipMemcpy(&st, d_leaf_stats_.ptr, sizeof(st), hipMemcpyDeviceToHost));
    st.cnt = static_cast<int>(global_cnt);
    HIP_OK(hipMemcpy(d_leaf_stats_.ptr, &st, sizeof(st), hipMemcpyHostToDevice));
  }

  BuildHistogram(leaf_slot_[0], leaf_begin_[0], leaf_cnt_[0]);
  ReduceHistogram(leaf_slot_[0]);
  LaunchBestSplit(0, leaf_slot_[0]);

  int num_leaves = 1;
  hipk::SplitRec winner;
  int winner_leaf = -1;
  for (int split_i = 0; split_i < nl - 1; ++split_i) {
    hipLaunchKernelGGL(hipk::k_best_overall, dim3(1), dim3(256), 0, stream_,
                       d_leaf_best_.ptr, num_leaves, d_winner_.ptr, d_winner_leaf_.ptr);
    HIP_OK(hipMemcpyAsync(&winner, d_winner_.ptr, sizeof(winner), hipMemcpyDeviceToHost,
                          stream_));
    HIP_OK(hipMemcpyAsync(&winner_leaf, d_winner_leaf_.ptr, sizeof(int),
                          hipMemcpyDeviceToHost, stream_));
    HIP_OK(hipStreamSynchronize(stream_));
    if (winner_leaf < 0) break;

    const int L = winner_leaf;
    const int R = num_leaves;
    const int f = winner.feature;
    const BinMapper* mapper = train_data_->FeatureBinMapper(f);
    const int orig_f = train_data_->RealFeatureIndex(f);

    // partition rows (exact left count comes back)
    const int left_exact = PartitionLeaf(L, winner, nullptr);
    const int parent_cnt = leaf_cnt_[L];
    const int right_exact = parent_cnt - left_exact;

    hipk::LeafStat parent;
    HIP_OK(hipMemcpy(&parent, d_leaf_stats_.ptr + L, sizeof(parent), hipMemcpyDeviceToHost));
    const double right_h = parent.sum_h - winner.left_h;

    // tree structure
    if (mapper->bin_type() == BinType::kCategorical) {
      // one-hot split: bin == winner.bin goes left
      const int cat = static_cast<int>(mapper->BinToValue(winner.bin));
      std::vector<uint32_t> bits(std::max(cat, 0) / 32 + 1, 0);
      if (cat >= 0) bits[cat >> 5] |= 1u << (cat & 31);
      tree->SplitCategorical(L, f, orig_f, bits.data(), static_cast<int>(bits.size()),
                             winner.left_out, winner.right_out, left_exact, right_exact,
                             winner.left_h, right_h, static_cast<float>(winner.gain),
                             mapper->missing_type());
    } else {
      tree->Split(L, f, orig_f, winner.bin, mapper->BinToValue(winner.bin), winner.left_out,
                  winner.right_out, left_exact, right_exact, winner.left_h, right_h,
                  static_cast<float>(winner.gain), mapper->missing_type(),
                  winner.default_left != 0);
    }

    // host partition bookkeeping
    leaf_begin_[R] = leaf_begin_[L] + left_exact;
    leaf_cnt_[R] = right_exact;
    leaf_cnt_[L] = left_exact;

    // child stats on device
    {
      hipk::LeafStat ls{winner.left_g, winner.left_h, left_exact, 0};
      hipk::LeafStat rs{parent.sum_g - winner.left_g, parent.sum_h - winner.left_h,
                        right_exact, 0};
      if (GpuComm::Get().active()) {
        // exact global child counts
        int64_t lc = left_exact;
        // reuse the small int64 buffer pattern
        static int64_t* d_lc = nullptr;
        if (!d_lc) HIP_OK(hipMalloc(&d_lc, sizeof(int64_t)));
        HIP_OK(hipMemcpy(d_lc, &lc, sizeof(int64_t), hipMemcpyHostToDevice));
        NCCL_OK(ncclAllReduce(d_lc, d_lc, 1, ncclInt64, ncclSum, GpuComm::Get().comm,
                              stream_));
        HIP_OK(hipStreamSynchronize(stream_));
        int64_t glc = 0;
        HIP_OK(hipMemcpy(&glc, d_lc, sizeof(int64_t), hipMemcpyDeviceToHost));
        ls.cnt = static_cast<int>(glc);
        rs.cnt = parent.cnt - ls.cnt;
      }
      HIP_OK(hipMemcpyAsync(d_leaf_stats_.ptr + L, &ls, sizeof(ls), hipMemcpyHostToDevice,
                            stream_));
      HIP_OK(hipMemcpyAsync(d_leaf_stats_.ptr + R, &rs, sizeof(rs), hipMemcpyHostToDevice,
                            stream_));
      // smaller/larger by (global) counts for the subtraction trick
      const int parent_slot = leaf_slot_[L];
      const int spare_slot = R;
      const bool left_smaller = ls.cnt <= rs.cnt;
      const int small_leaf = left_smaller ? L : R;
      const int large_leaf = left_smaller ? R : L;
      leaf_slot_[small_leaf] = spare_slot;
      leaf_slot_[large_leaf] = parent_slot;
      ++num_leaves;
      BuildHistogram(spare_slot, leaf_begin_[small_leaf], leaf_cnt_[small_leaf]);
      ReduceHistogram(spare_slot);
      const int n_elem = total_bins_ * 2;
      hipLaunchKernelGGL(hipk::k_hist_subtract, dim3((n_elem + 255) / 256), dim3(256), 0,
                         stream_, d_hist_.ptr + static_cast<size_t>(parent_slot) * n_elem,
                         d_hist_.ptr + static_cast<size_t>(spare_slot) * n_elem, n_elem);
      LaunchBestSplit(small_leaf, spare_slot);
      LaunchBestSplit(large_leaf, parent_slot);
    }
  }
  HIP_OK(hipStreamSynchronize(stream_));
  return tree.release();
}

void HIPTreeLearner::AddPredictionToScore(const Tree* tree, double* /*out_score*/) {
  // device score update over the final partition
  const int nl = tree->num_leaves();
  // leaf segments are nested, not ordered by leaf id: sort by begin for binary search
  std::vector<int> order(nl);
  for (int l = 0; l < nl; ++l) order[l] = l;
  std::sort(order.begin(), order.end(),
            [&](int a, int b) { return leaf_begin_[a] < leaf_begin_[b]; });
  std::vector<double> outs(nl);
  std::vector<int> sorted_begin(nl), sorted_cnt(nl);
  for (int k = 0; k < nl; ++k) {
    outs[k] = tree->LeafOutput(order[k]);
    sorted_begin[k] = leaf_begin_[order[k]];
    sorted_cnt[k] = leaf_cnt_[order[k]];
  }
  HIP_OK(hipMemcpyAsync(d_leaf_out_.ptr, outs.data(), sizeof(double) * nl,
                        hipMemcpyHostToDevice, stream_));
  HIP_OK(hipMemcpyAsync(d_leaf_begin_.ptr, sorted_begin.data(), sizeof(int) * nl,
                        hipMemcpyHostToDevice, stream_));
  HIP_OK(hipMemcpyAsync(d_leaf_cnt_.ptr, sorted_cnt.data(), sizeof(int) * nl,
                        hipMemcpyHostToDevice, stream_));
  const int n = static_cast<int>(used_cnt_);
  hipLaunchKernelGGL(hipk::k_score_update, dim3((n + 255) / 256), dim3(256), 0, stream_,
                     d_idx_.ptr, d_leaf_begin_.ptr, d_leaf_cnt_.ptr, nl, n, d_leaf_out_.ptr,
                     d_score_);
  // out-of-bag rows via device tree walk
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
    std::vector<int> feat(ni), thr(ni), lc(ni), rc(ni), nb(ni);
    std::vector<uint8_t> dl(ni);
    for (int i2 = 0; i2 < ni; ++i2) {
      feat[i2] = tree->split_feature_inner(i2);
      thr[i2] = static_cast<int>(tree->threshold_in_bin(i2));
      lc[i2] = tree->left_child(i2);
      rc[i2] = tree->right_child(i2);
      nb[i2] = feat_meta_host_[feat[i2]].nan_bin;
      dl[i2] = (tree->decision_type(i2) & Tree::kDefaultLeftMask) ? 1 : 0;
    }
    HIP_OK(hipMemcpyAsync(d_tw_feat_.ptr, feat.data(), sizeof(int) * ni,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_tw_thr_.ptr, thr.data(), sizeof(int) * ni, hipMemcpyHostToDevice,
                          stream_));
    HIP_OK(hipMemcpyAsync(d_tw_left_.ptr, lc.data(), sizeof(int) * ni, hipMemcpyHostToDevice,
                          stream_));
    HIP_OK(hipMemcpyAsync(d_tw_right_.ptr, rc.data(), sizeof(int) * ni,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_tw_nan_.ptr, nb.data(), sizeof(int) * ni, hipMemcpyHostToDevice,
                          stream_));
    HIP_OK(hipMemcpyAsync(d_tw_dl_.ptr, dl.data(), ni, hipMemcpyHostToDevice, stream_));
    std::vector<double> out_by_leaf(nl);
    for (int l = 0; l < nl; ++l) out_by_leaf[l] = tree->LeafOutput(l);
    HIP_OK(hipMemcpyAsync(d_tw_out_.ptr, out_by_leaf.data(), sizeof(double) * nl,
                          hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_oob_.ptr, oob.data(), sizeof(uint32_t) * oob.size(),
                          hipMemcpyHostToDevice, stream_));
    hipLaunchKernelGGL(hipk::k_tree_predict_add,
                       dim3((static_cast<int>(oob.size()) + 255) / 256), dim3(256), 0,
                       stream_, (const uint8_t* const*)d_col_ptrs_.ptr, d_tw_feat_.ptr,
                       d_tw_thr_.ptr, d_tw_left_.ptr, d_tw_right_.ptr, d_tw_nan_.ptr,
                       d_tw_dl_.ptr, d_tw_out_.ptr, d_oob_.ptr,
                       static_cast<int>(oob.size()), d_score_);
  }
}

void HIPTreeLearner::RenewTreeOutput(Tree* tree, const ObjectiveFunction* obj,
                                     std::function<double(const label_t*, int)>, data_size_t,
                                     const data_size_t*, data_size_t,
                                     const double* train_score) {
  if (obj == nullptr || !obj->NeedRenewTreeOutput()) return;
  // correctness path: download partition indices + scores, renew on host
  std::vector<uint32_t> idx(used_cnt_);
  HIP_OK(hipStreamSynchronize(stream_));
  HIP_OK(hipMemcpy(idx.data(), d_idx_.ptr, sizeof(uint32_t) * used_cnt_,
                   hipMemcpyDeviceToHost));
  std::vector<double> score(num_data_);
  HIP_OK(hipMemcpy(score.data(), d_score_.ptr, sizeof(double) * num_data_,
                   hipMemcpyDeviceToHost));
  (void)train_score;
  const int nl = tree->num_leaves();
  for (int l = 0; l < nl; ++l) {
    if (leaf_cnt_[l] == 0) continue;
    std::vector<data_size_t> rows(leaf_cnt_[l]);
    for (int i = 0; i < leaf_cnt_[l]; ++i)
      rows[i] = static_cast<data_size_t>(idx[leaf_begin_[l] + i]);
    tree->SetLeafOutput(l, obj->RenewTreeOutput(tree->LeafOutput(l), rows.data(),
                                                leaf_cnt_[l], score.data()));
  }
}

namespace hipk {
__global__ void k_iota(uint32_t* p, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = static_cast<uint32_t>(i);
}
}  // namespace hipk

// ------------------------------------------------------------------ registration
namespace {
TreeLearner* CreateHIP(const Config* cfg) { return new HIPTreeLearner(cfg); }
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
  } catch (...) {
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
