/*!
 * migbm BinMapper — per-feature value->bin discretization.
 * Capability parity target: reference include/LightGBM/bin.h + src/io/bin.cpp
 * (BinMapper::FindBin, ValueToBin, categorical dictionaries, missing handling).
 * Fresh implementation tuned for a GPU-resident row-major bin matrix: every bin
 * (including the most frequent one) is materialized, so no FixHistogram pass is
 * needed on device.
 */
#ifndef MIGBM_BIN_H_
#define MIGBM_BIN_H_

#include "common.h"

#include <string>
#include <unordered_map>
#include <vector>

namespace migbm {

enum class BinType : int8_t { kNumerical = 0, kCategorical = 1 };
enum class MissingType : int8_t { kNone = 0, kZero = 1, kNaN = 2 };

class BinMapper {
 public:
  BinMapper() = default;

  /*! Find bin boundaries from sampled values.
   *  \param values sampled non-trivial values (may contain NaN), modified in place
   *  \param num_sample_values count of sampled values
   *  \param total_sample_cnt total rows sampled (>= num_sample_values; difference = zeros)
   */
  void FindBin(double* values, int num_sample_values, size_t total_sample_cnt, int max_bin,
               int min_data_in_bin, int min_split_data, bool pre_filter, BinType bin_type,
               bool use_missing, bool zero_as_missing,
               const std::vector<double>* forced_bounds = nullptr);

  /*! Map a raw value to its bin. */
  inline uint32_t ValueToBin(double value) const {
    if (bin_type_ == BinType::kCategorical) {
      // NaN / negative / unseen categories all land in bin 0 (the reference's
      // dummy NaN bin, bin_2_categorical_[0] == -1) — DISTINCT from category 0
      if (std::isnan(value)) return 0;
      int cat = static_cast<int>(value);
      if (cat < 0) return 0;
      auto it = categorical_2_bin_.find(cat);
      if (it == categorical_2_bin_.end()) return 0;
      return it->second;
    }
    if (std::isnan(value)) {
      if (missing_type_ == MissingType::kNaN) return static_cast<uint32_t>(num_bin_ - 1);
      value = 0.0;
    }
    if (missing_type_ == MissingType::kZero && value == 0.0)
      return static_cast<uint32_t>(num_bin_ - 1);
    int lo = 0, hi = num_numeric_bin_ - 1;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (value <= bin_upper_bound_[mid]) hi = mid;
      else lo = mid + 1;
    }
    return static_cast<uint32_t>(lo);
  }

  /*! Representative value for a bin (upper boundary for numerical). */
  double BinToValue(uint32_t bin) const {
    if (bin_type_ == BinType::kCategorical) {
      return bin < bin_2_categorical_.size() ? bin_2_categorical_[bin] : -1.0;
    }
    if (static_cast<int>(bin) >= num_numeric_bin_) return std::numeric_limits<double>::quiet_NaN();
    return bin_upper_bound_[bin];
  }

  int num_bin() const { return num_bin_; }
  bool is_trivial() const { return is_trivial_; }
  BinType bin_type() const { return bin_type_; }
  MissingType missing_type() const { return missing_type_; }
  uint32_t most_freq_bin() const { return most_freq_bin_; }
  uint32_t default_bin() const { return default_bin_; }
  double sparse_rate() const { return sparse_rate_; }
  double min_value() const { return min_val_; }
  double max_value() const { return max_val_; }
  const std::vector<double>& bin_upper_bound() const { return bin_upper_bound_; }
  const std::vector<int>& bin_2_categorical() const { return bin_2_categorical_; }
  int num_numeric_bin() const { return num_numeric_bin_; }
  /*! bin that NaN maps to (num_bin-1 when missing_type==NaN/Zero), or -1 */
  int nan_bin() const { return missing_type_ != MissingType::kNone ? num_bin_ - 1 : -1; }

  int CategoryToBin(int cat) const {
    auto it = categorical_2_bin_.find(cat);
    return it == categorical_2_bin_.end() ? -1 : static_cast<int>(it->second);
  }

  /*! model text "feature_infos" entry: "[min:max]" numerical, "cat1:cat2:..." categorical,
   *  "none" trivial. Matches the reference format (gbdt_model_text.cpp). */
  std::string ToFeatureInfoString() const;

  std::string ToString() const;           // full serialization (dataset binary file)
  void FromString(const std::string& s);

 private:
  friend class Dataset;
  int num_bin_ = 1;
  int num_numeric_bin_ = 1;               // numeric bins; NaN bin (if any) appended after
  bool is_trivial_ = true;
  BinType bin_type_ = BinType::kNumerical;
  MissingType missing_type_ = MissingType::kNone;
  uint32_t most_freq_bin_ = 0;
  uint32_t default_bin_ = 0;
  double sparse_rate_ = 0.0;
  double min_val_ = 0.0, max_val_ = 0.0;
  std::vector<double> bin_upper_bound_;   // ascending; last numeric bound = +inf
  std::vector<int> bin_2_categorical_;
  std::unordered_map<int, uint32_t> categorical_2_bin_;
};

}  // namespace migbm

#endif  // MIGBM_BIN_H_
