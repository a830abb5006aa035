/*! migbm gain-scan implementation (CPU oracle shared by learners and HIP-kernel tests).
 *  Numerics parity: reference feature_histogram.hpp FindBestThresholdSequentially /
 *  categorical scan — formulas re-derived from the published algorithm. */
#include "migbm/feature_histogram.h"

#include <algorithm>

namespace migbm {

void FindBestThresholdNumerical(const hist_t* hist, int num_bin, int num_numeric_bin,
                                int nan_bin, const LeafContext& leaf, const Config& cfg,
                                int8_t monotone_constraint, int rand_threshold,
                                SplitInfo* out, const MonoAdvBounds* adv) {
  const double l1 = cfg.lambda_l1, l2 = cfg.lambda_l2;
  const double mds = cfg.max_delta_step, smooth = cfg.path_smooth;
  const double min_hess = cfg.min_sum_hessian_in_leaf;
  const data_size_t min_cnt = cfg.min_data_in_leaf;
  const double cnt_factor = leaf.num_data > 0 && leaf.sum_hessian > 0
                                ? static_cast<double>(leaf.num_data) / leaf.sum_hessian
                                : 1.0;
  const double parent_gain = GainMath::GetLeafGain(leaf.sum_gradient, leaf.sum_hessian, l1, l2,
                                                   mds, smooth, leaf.num_data,
                                                   leaf.parent_output);
  const double min_gain_shift = parent_gain + cfg.min_gain_to_split;

  const bool has_nan = nan_bin >= 0 && nan_bin < num_bin;
  double g_nan = 0.0, h_nan = 0.0;
  if (has_nan) { g_nan = hist[2 * nan_bin]; h_nan = hist[2 * nan_bin + 1]; }

  double best_gain = kMinScore;
  int best_t = -1;
  bool best_default_left = false;
  double best_lg = 0, best_lh = 0;
  double best_llo = leaf.out_lo, best_lhi = leaf.out_hi;
  double best_rlo = leaf.out_lo, best_rhi = leaf.out_hi;

  // advanced monotone: suffix extrema so right-child bounds over bins [t+1..end)
  // are O(1) per threshold; left-child extrema run forward with the scan
  std::vector<double> rmax_lo, rmin_hi;
  if (adv != nullptr && !adv->empty()) {
    rmax_lo.resize(num_numeric_bin + 1);
    rmin_hi.resize(num_numeric_bin + 1);
    rmax_lo[num_numeric_bin] = -std::numeric_limits<double>::infinity();
    rmin_hi[num_numeric_bin] = std::numeric_limits<double>::infinity();
    for (int b = num_numeric_bin - 1; b >= 0; --b) {
      rmax_lo[b] = std::max(rmax_lo[b + 1], adv->lo[b]);
      rmin_hi[b] = std::min(rmin_hi[b + 1], adv->hi[b]);
    }
  }
  double run_lo = -std::numeric_limits<double>::infinity();
  double run_hi = std::numeric_limits<double>::infinity();

  // prefix over numeric bins: for threshold t, left(no-missing) = sum bins 0..t.
  // Without missing the last numeric bin can't be a threshold; WITH a NaN bin it
  // can — left = all numeric values, right = missing only (the reference's
  // everything-vs-NaN split, required e.g. for constant columns with NaNs)
  double gl = 0.0, hl = 0.0;
  const int t_max = has_nan ? num_numeric_bin - 1 : num_numeric_bin - 2;
  for (int t = 0; t <= t_max; ++t) {
    gl += hist[2 * t];
    hl += hist[2 * t + 1];
    double left_lo = leaf.out_lo, left_hi = leaf.out_hi;
    double right_lo = leaf.out_lo, right_hi = leaf.out_hi;
    if (!rmax_lo.empty()) {
      run_lo = std::max(run_lo, adv->lo[t]);
      run_hi = std::min(run_hi, adv->hi[t]);
      left_lo = run_lo; left_hi = run_hi;
      right_lo = rmax_lo[t + 1]; right_hi = rmin_hi[t + 1];
    }
    if (rand_threshold >= 0 && t != rand_threshold) continue;
    // two missing placements (if missing exists); else single evaluation
    const int n_variants = has_nan ? 2 : 1;
    for (int v = 0; v < n_variants; ++v) {
      const bool missing_left = (v == 1);
      double sgl = gl + (missing_left ? g_nan : 0.0);
      double shl = hl + (missing_left ? h_nan : 0.0);
      double sgr = leaf.sum_gradient - sgl;
      double shr = leaf.sum_hessian - shl;
      data_size_t lc = static_cast<data_size_t>(Common::RoundInt(shl * cnt_factor));
      data_size_t rc = leaf.num_data - lc;
      if (shl < min_hess || lc < min_cnt) continue;
      if (shr < min_hess || rc < min_cnt) break;  // further t only shrinks the right side for v==0; for v==1 approximately too
      double lo = GainMath::CalculateSplittedLeafOutput(sgl, shl, l1, l2, mds, smooth, lc,
                                                        leaf.parent_output);
      double ro = GainMath::CalculateSplittedLeafOutput(sgr, shr, l1, l2, mds, smooth, rc,
                                                        leaf.parent_output);
      // clamp candidate outputs into the inherited bounds (basic/intermediate:
      // one scalar pair per leaf; advanced: per-threshold piecewise bounds)
      lo = std::min(std::max(lo, left_lo), left_hi);
      ro = std::min(std::max(ro, right_lo), right_hi);
      if (monotone_constraint != 0) {
        if (monotone_constraint > 0 && lo > ro) continue;
        if (monotone_constraint < 0 && lo < ro) continue;
      }
      double gain = GainMath::GetLeafGainGivenOutput(sgl, shl, l1, l2, lo) +
                    GainMath::GetLeafGainGivenOutput(sgr, shr, l1, l2, ro);
      if (gain <= min_gain_shift) continue;
      if (gain > best_gain) {
        best_gain = gain;
        best_t = t;
        best_default_left = missing_left;
        best_lg = sgl; best_lh = shl;
        best_llo = left_lo; best_lhi = left_hi;
        best_rlo = right_lo; best_rhi = right_hi;
      }
    }
  }
  if (best_t < 0) { out->gain = kMinScore; return; }
  out->threshold = static_cast<uint32_t>(best_t);
  out->default_left = best_default_left;
  out->gain = best_gain - min_gain_shift + cfg.min_gain_to_split;  // report gain above parent
  out->left_sum_gradient = best_lg;
  out->left_sum_hessian = best_lh;
  out->right_sum_gradient = leaf.sum_gradient - best_lg;
  out->right_sum_hessian = leaf.sum_hessian - best_lh;
  out->left_count = static_cast<data_size_t>(Common::RoundInt(best_lh * cnt_factor));
  out->right_count = leaf.num_data - out->left_count;
  out->left_output = GainMath::CalculateSplittedLeafOutput(best_lg, best_lh, l1, l2, mds, smooth,
                                                           out->left_count, leaf.parent_output);
  out->right_output = GainMath::CalculateSplittedLeafOutput(
      out->right_sum_gradient, out->right_sum_hessian, l1, l2, mds, smooth, out->right_count,
      leaf.parent_output);
  out->left_output = std::min(std::max(out->left_output, best_llo), best_lhi);
  out->right_output = std::min(std::max(out->right_output, best_rlo), best_rhi);
  out->monotone_type = monotone_constraint;
}

void FindBestThresholdCategorical(const hist_t* hist, int num_bin, const LeafContext& leaf,
                                  const Config& cfg, SplitInfo* out) {
  const double l1 = cfg.lambda_l1;
  const double l2 = cfg.lambda_l2 + cfg.cat_l2;
  const double min_hess = cfg.min_sum_hessian_in_leaf;
  const data_size_t min_cnt = cfg.min_data_in_leaf;
  const double cnt_factor = leaf.num_data > 0 && leaf.sum_hessian > 0
                                ? static_cast<double>(leaf.num_data) / leaf.sum_hessian
                                : 1.0;
  const double parent_gain = GainMath::GetLeafGain(leaf.sum_gradient, leaf.sum_hessian, l1,
                                                   cfg.lambda_l2, cfg.max_delta_step,
                                                   cfg.path_smooth, leaf.num_data,
                                                   leaf.parent_output);
  const double min_gain_shift = parent_gain + cfg.min_gain_to_split;

  double best_gain = kMinScore;
  std::vector<int> best_cats;
  double best_lg = 0, best_lh = 0;

  auto eval_subset = [&](const std::vector<int>& bins_in, double sgl, double shl) {
    double sgr = leaf.sum_gradient - sgl;
    double shr = leaf.sum_hessian - shl;
    data_size_t lc = static_cast<data_size_t>(Common::RoundInt(shl * cnt_factor));
    data_size_t rc = leaf.num_data - lc;
    if (shl < min_hess || lc < min_cnt) return;
    if (shr < min_hess || rc < min_cnt || rc < cfg.min_data_per_group) return;
    double gain = GainMath::GetSplitGains(sgl, shl, sgr, shr, l1, l2, cfg.max_delta_step, 0.0,
                                          lc, rc, leaf.parent_output);
    if (gain <= min_gain_shift) return;
    if (gain > best_gain) {
      best_gain = gain;
      best_cats = bins_in;
      best_lg = sgl; best_lh = shl;
    }
  };

  if (num_bin <= cfg.max_cat_to_onehot) {
    // one-hot: left = single bin
    for (int b = 0; b < num_bin; ++b) {
      eval_subset({b}, hist[2 * b], hist[2 * b + 1]);
    }
  } else {
    // sorted scan by grad/(hess + cat_smooth), both ends, up to max_cat_threshold
    std::vector<int> order;
    for (int b = 0; b < num_bin; ++b) {
      double cnt_b = hist[2 * b + 1] * cnt_factor;
      if (cnt_b >= cfg.cat_smooth) order.push_back(b);
    }
    std::sort(order.begin(), order.end(), [&](int a, int b) {
      double ra = hist[2 * a] / (hist[2 * a + 1] + cfg.cat_smooth);
      double rb = hist[2 * b] / (hist[2 * b + 1] + cfg.cat_smooth);
      return ra < rb;
    });
    const int n = static_cast<int>(order.size());
    const int limit = std::min(cfg.max_cat_threshold, n - 1);
    for (int dir = 0; dir < 2; ++dir) {
      double sgl = 0, shl = 0;
      data_size_t cnt_cur_group = 0;
      std::vector<int> taken;
      for (int k = 0; k < limit; ++k) {
        int b = dir == 0 ? order[k] : order[n - 1 - k];
        sgl += hist[2 * b];
        shl += hist[2 * b + 1];
        cnt_cur_group += static_cast<data_size_t>(
            Common::RoundInt(hist[2 * b + 1] * cnt_factor));
        taken.push_back(b);
        // group gating (parity: reference min_data_per_group): only evaluate a
        // boundary once the accumulated group since the last evaluation is big
        // enough — prevents tiny-category overfitting
        if (cnt_cur_group < cfg.min_data_per_group) continue;
        cnt_cur_group = 0;
        eval_subset(taken, sgl, shl);
      }
    }
  }
  if (best_cats.empty()) { out->gain = kMinScore; return; }
  // normalize: the NaN/unseen bin 0 must route RIGHT. The raw-value predictor
  // sends NaN/unseen right unconditionally (category bitsets cannot express the
  // dummy category -1), so a left-set containing bin 0 would diverge from
  // training. A categorical split is symmetric: complement the subset instead.
  if (std::find(best_cats.begin(), best_cats.end(), 0) != best_cats.end()) {
    std::vector<int> comp;
    comp.reserve(num_bin - best_cats.size());
    std::vector<uint8_t> in(num_bin, 0);
    for (int b : best_cats) in[b] = 1;
    for (int b = 0; b < num_bin; ++b)
      if (!in[b]) comp.push_back(b);
    if (comp.empty()) { out->gain = kMinScore; return; }
    best_cats = std::move(comp);
    best_lg = leaf.sum_gradient - best_lg;
    best_lh = leaf.sum_hessian - best_lh;
  }
  out->gain = best_gain - min_gain_shift + cfg.min_gain_to_split;
  out->left_sum_gradient = best_lg;
  out->left_sum_hessian = best_lh;
  out->right_sum_gradient = leaf.sum_gradient - best_lg;
  out->right_sum_hessian = leaf.sum_hessian - best_lh;
  out->left_count = static_cast<data_size_t>(Common::RoundInt(best_lh * cnt_factor));
  out->right_count = leaf.num_data - out->left_count;
  out->left_output = GainMath::CalculateSplittedLeafOutput(best_lg, best_lh, l1, l2,
                                                           cfg.max_delta_step);
  out->right_output = GainMath::CalculateSplittedLeafOutput(
      out->right_sum_gradient, out->right_sum_hessian, l1, l2, cfg.max_delta_step);
  out->default_left = false;
  // bin-level bitset
  int max_b = *std::max_element(best_cats.begin(), best_cats.end());
  out->cat_bitset_inner.assign(max_b / 32 + 1, 0);
  for (int b : best_cats) out->cat_bitset_inner[b >> 5] |= (1u << (b & 31));
  out->threshold = static_cast<uint32_t>(best_cats.size());
}

}  // namespace migbm
