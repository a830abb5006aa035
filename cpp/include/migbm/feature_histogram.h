/*!
 * migbm split gain math — best-threshold search over (grad,hess) histograms.
 * Numerics parity target: reference src/treelearner/feature_histogram.hpp
 * (CalculateSplittedLeafOutput / GetSplitGains / FindBestThresholdSequentially semantics,
 * including L1 thresholding, max_delta_step, path smoothing, hessian-derived counts).
 * Free functions so the CPU learner, the data-parallel learner and the HIP-kernel unit
 * tests share one oracle.
 */
#ifndef MIGBM_FEATURE_HISTOGRAM_H_
#define MIGBM_FEATURE_HISTOGRAM_H_

#include "common.h"
#include "config.h"

#include <vector>

namespace migbm {

struct SplitInfo {
  int feature = -1;                 // inner feature index
  uint32_t threshold = 0;           // bin threshold (numerical) or #cats (categorical)
  double left_output = 0.0, right_output = 0.0;
  double gain = kMinScore;
  double left_sum_gradient = 0.0, left_sum_hessian = 0.0;
  double right_sum_gradient = 0.0, right_sum_hessian = 0.0;
  data_size_t left_count = 0, right_count = 0;
  bool default_left = true;
  int8_t monotone_type = 0;
  std::vector<uint32_t> cat_bitset;  // non-empty => categorical split (bin-level bitset)
  std::vector<uint32_t> cat_bitset_inner;

  bool IsValid() const { return gain > kMinScore; }
  void Reset() { *this = SplitInfo(); }

  /*! total order for distributed argmax ties (parity: split_info.hpp operator>) */
  bool operator>(const SplitInfo& other) const {
    double g1 = IsValid() ? gain : kMinScore;
    double g2 = other.IsValid() ? other.gain : kMinScore;
    if (g1 != g2) return g1 > g2;
    if (feature != other.feature) return feature < other.feature;  // tie: smaller feature wins
    return false;
  }
};

/*! Gain/output math (all static, shared CPU oracle). */
struct GainMath {
  /*! leaf output = -ThresholdL1(G,l1)/(H+l2), optionally clipped by max_delta_step,
   *  optionally path-smoothed. */
  static inline double CalculateSplittedLeafOutput(double sum_g, double sum_h, double l1,
                                                   double l2, double max_delta_step) {
    double ret = -Common::ThresholdL1(sum_g, l1) / (sum_h + l2);
    if (max_delta_step <= 0.0 || std::fabs(ret) <= max_delta_step) return ret;
    return ret > 0 ? max_delta_step : -max_delta_step;
  }
  static inline double CalculateSplittedLeafOutput(double sum_g, double sum_h, double l1,
                                                   double l2, double max_delta_step,
                                                   double smoothing, data_size_t num_data,
                                                   double parent_output) {
    double ret = CalculateSplittedLeafOutput(sum_g, sum_h, l1, l2, max_delta_step);
    if (smoothing <= 0.0) return ret;
    double n = static_cast<double>(num_data);
    return ret * (n / smoothing) / (n / smoothing + 1.0) +
           parent_output / (n / smoothing + 1.0);
  }
  static inline double GetLeafGainGivenOutput(double sum_g, double sum_h, double l1, double l2,
                                              double output) {
    const double sg_l1 = Common::ThresholdL1(sum_g, l1);
    return -(2.0 * sg_l1 * output + (sum_h + l2) * output * output);
  }
  static inline double GetLeafGain(double sum_g, double sum_h, double l1, double l2,
                                   double max_delta_step, double smoothing,
                                   data_size_t num_data, double parent_output) {
    if (max_delta_step <= 0.0 && smoothing <= 0.0) {
      const double sg_l1 = Common::ThresholdL1(sum_g, l1);
      return (sg_l1 * sg_l1) / (sum_h + l2);
    }
    const double output = CalculateSplittedLeafOutput(sum_g, sum_h, l1, l2, max_delta_step,
                                                      smoothing, num_data, parent_output);
    return GetLeafGainGivenOutput(sum_g, sum_h, l1, l2, output);
  }
  static inline double GetSplitGains(double sum_left_g, double sum_left_h, double sum_right_g,
                                     double sum_right_h, double l1, double l2,
                                     double max_delta_step, double smoothing,
                                     data_size_t left_count, data_size_t right_count,
                                     double parent_output) {
    return GetLeafGain(sum_left_g, sum_left_h, l1, l2, max_delta_step, smoothing, left_count,
                       parent_output) +
           GetLeafGain(sum_right_g, sum_right_h, l1, l2, max_delta_step, smoothing, right_count,
                       parent_output);
  }
};

/*! Context for a gain scan of one leaf. */
struct LeafContext {
  double sum_gradient = 0.0;
  double sum_hessian = 0.0;
  data_size_t num_data = 0;
  double parent_output = 0.0;   // for path smoothing
  int depth = 0;
  // monotone-constraint output bounds (BasicLeafConstraints propagation)
  double out_lo = -std::numeric_limits<double>::infinity();
  double out_hi = std::numeric_limits<double>::infinity();
};

/*! Per-bin monotone output bounds for monotone_constraints_method=advanced
 *  ("monotone precise" mode; parity: reference AdvancedLeafConstraints /
 *  CumulativeFeatureConstraint, monotone_constraints.hpp:143-289 — dense per-bin
 *  representation instead of the reference's threshold-interval arrays).
 *  lo[b]/hi[b] bound the output of any child leaf whose feature range covers
 *  bin b; a candidate child spanning bins [a..c] must lie in
 *  [max(lo[a..c]), min(hi[a..c])]. */
struct MonoAdvBounds {
  std::vector<double> lo, hi;  // size num_numeric_bin
  bool empty() const { return lo.empty(); }
};

/*! Best numerical threshold for one feature.
 *  hist: (g,h) pairs, num_bin entries; nan_bin: index of the NaN/Zero bin or -1.
 *  num_numeric_bin: bins eligible as thresholds (nan bin excluded).
 *  rand_threshold: if >=0 (extra_trees), only this bin index is evaluated.
 *  adv: per-threshold monotone bounds (advanced mode) — when non-null they
 *  REPLACE the scalar leaf bounds for left/right output clamping. */
void FindBestThresholdNumerical(const hist_t* hist, int num_bin, int num_numeric_bin,
                                int nan_bin, const LeafContext& leaf, const Config& cfg,
                                int8_t monotone_constraint, int rand_threshold,
                                SplitInfo* out, const MonoAdvBounds* adv = nullptr);

/*! Best categorical split (one-hot or sorted-subset scan).
 *  Emits a bin-level bitset in out->cat_bitset_inner. */
void FindBestThresholdCategorical(const hist_t* hist, int num_bin, const LeafContext& leaf,
                                  const Config& cfg, SplitInfo* out);

}  // namespace migbm

#endif  // MIGBM_FEATURE_HISTOGRAM_H_
