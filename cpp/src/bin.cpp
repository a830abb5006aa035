/*! migbm BinMapper implementation.
 *  Equal-frequency greedy binning from a value sample; categorical dictionary binning by
 *  descending frequency. Capability target: reference src/io/bin.cpp (FindBin /
 *  GreedyFindBin semantics — re-derived, not copied).
 */
#include "migbm/bin.h"

#include <algorithm>
#include <cmath>

namespace migbm {

namespace {

/*! Midpoint boundary between two adjacent distinct values (numerically safe). */
inline double MidPoint(double a, double b) {
  double m = a + (b - a) / 2.0;
  if (!(m > a && m <= b)) m = b;  // degenerate spacing: fall back to upper value
  return m;
}

/*! Greedy equal-frequency binning over (distinct value, count) pairs.
 *  Returns ascending upper bounds; final bound is +inf. */
std::vector<double> GreedyFindBin(const std::vector<double>& dv, const std::vector<int>& cnt,
                                  int max_bin, size_t total_cnt, int min_data_in_bin) {
  std::vector<double> bounds;
  const int n = static_cast<int>(dv.size());
  if (n == 0) return bounds;
  if (n <= max_bin) {
    // one bin per distinct value, but respect min_data_in_bin by merging tiny bins
    int i = 0;
    int acc = 0;
    for (i = 0; i < n; ++i) {
      acc += cnt[i];
      if (acc >= min_data_in_bin || i == n - 1) {
        if (i < n - 1) bounds.push_back(MidPoint(dv[i], dv[i + 1]));
        acc = 0;
      }
    }
    if (bounds.empty() || bounds.back() != std::numeric_limits<double>::infinity())
      bounds.push_back(std::numeric_limits<double>::infinity());
    return bounds;
  }
  // equal-frequency with per-bin minimum
  double mean_per_bin = static_cast<double>(total_cnt) / max_bin;
  double target = std::max<double>(mean_per_bin, min_data_in_bin);
  double acc = 0.0;
  size_t rest = total_cnt;
  int rest_bins = max_bin;
  for (int i = 0; i < n; ++i) {
    acc += cnt[i];
    rest -= cnt[i];
    bool last_value = (i == n - 1);
    if (!last_value && acc >= target && rest_bins > 1) {
      bounds.push_back(MidPoint(dv[i], dv[i + 1]));
      acc = 0.0;
      --rest_bins;
      if (rest_bins > 0) target = std::max<double>(static_cast<double>(rest) / rest_bins,
                                                   min_data_in_bin);
    }
  }
  bounds.push_back(std::numeric_limits<double>::infinity());
  return bounds;
}

}  // namespace

void BinMapper::FindBin(double* values, int num_sample_values, size_t total_sample_cnt,
                        int max_bin, int min_data_in_bin, int /*min_split_data*/,
                        bool pre_filter, BinType bin_type, bool use_missing,
                        bool zero_as_missing, const std::vector<double>* forced_bounds) {
  bin_type_ = bin_type;
  // split NaN out
  int na_cnt = 0;
  int n = 0;
  for (int i = 0; i < num_sample_values; ++i) {
    if (std::isnan(values[i])) ++na_cnt;
    else values[n++] = values[i];
  }
  size_t zero_cnt = total_sample_cnt - static_cast<size_t>(n) - na_cnt;

  if (!use_missing) {
    missing_type_ = MissingType::kNone;
  } else if (zero_as_missing) {
    missing_type_ = MissingType::kZero;
  } else if (na_cnt > 0) {
    missing_type_ = MissingType::kNaN;
  } else {
    missing_type_ = MissingType::kNone;
  }

  if (bin_type_ == BinType::kCategorical) {
    // frequency-ordered category dictionary
    std::unordered_map<int, int> counts;
    for (int i = 0; i < n; ++i) {
      double v = values[i];
      if (v < 0) { ++na_cnt; continue; }  // negative categories treated as missing (ref behavior)
      counts[static_cast<int>(v)]++;
    }
    if (zero_cnt > 0) counts[0] += static_cast<int>(zero_cnt);
    std::vector<std::pair<int, int>> sorted(counts.begin(), counts.end());
    std::sort(sorted.begin(), sorted.end(), [](auto& a, auto& b) {
      return a.second > b.second || (a.second == b.second && a.first < b.first);
    });
    // cap at max_bin-1 categories (bin 0 reserved for unseen/other)
    int cap = std::min<int>(static_cast<int>(sorted.size()), max_bin - 1);
    // drop ultra-rare cats (<1% of sample) like the reference's cut by count
    bin_2_categorical_.clear();
    categorical_2_bin_.clear();
    bin_2_categorical_.push_back(-1);  // bin 0: other/unseen
    for (int i = 0; i < cap; ++i) {
      categorical_2_bin_[sorted[i].first] = static_cast<uint32_t>(i + 1);
      bin_2_categorical_.push_back(sorted[i].first);
    }
    num_bin_ = static_cast<int>(bin_2_categorical_.size());
    num_numeric_bin_ = num_bin_;
    missing_type_ = MissingType::kNone;  // unseen categories -> bin 0
    // a populated NaN/unseen bin 0 counts as a distinct value: a feature that is
    // one category plus NaNs is informative (reference test_categorical_handle_na)
    is_trivial_ = pre_filter &&
                  static_cast<int>(sorted.size()) + (na_cnt > 0 ? 1 : 0) <= 1;
    most_freq_bin_ = num_bin_ > 1 ? 1 : 0;
    default_bin_ = 0;
    sparse_rate_ = 0.0;
    return;
  }

  // numerical
  std::sort(values, values + n);
  min_val_ = n > 0 ? std::min(values[0], 0.0) : 0.0;
  max_val_ = n > 0 ? std::max(values[n - 1], 0.0) : 0.0;
  if (zero_cnt == 0 && n > 0) { min_val_ = values[0]; max_val_ = values[n - 1]; }

  // distinct values with counts; implicit zeros (rows not in the sample's value list)
  // folded into the zero entry at its sorted position
  std::vector<double> dv;
  std::vector<int> cnt;
  for (int i = 0; i < n; ++i) {
    if (!dv.empty() && values[i] == dv.back()) cnt.back()++;
    else { dv.push_back(values[i]); cnt.push_back(1); }
  }
  if (zero_cnt > 0) {
    auto it = std::lower_bound(dv.begin(), dv.end(), 0.0);
    size_t pos = static_cast<size_t>(it - dv.begin());
    if (it != dv.end() && *it == 0.0) {
      cnt[pos] += static_cast<int>(zero_cnt);
    } else {
      dv.insert(it, 0.0);
      cnt.insert(cnt.begin() + pos, static_cast<int>(zero_cnt));
    }
  }

  int usable_bins = max_bin;
  if (missing_type_ == MissingType::kNaN || missing_type_ == MissingType::kZero) usable_bins -= 1;
  if (missing_type_ == MissingType::kZero) {
    // remove zero from distinct list; zeros map to the NaN-equivalent bin
    for (size_t i = 0; i < dv.size(); ++i) {
      if (dv[i] == 0.0) { dv.erase(dv.begin() + i); cnt.erase(cnt.begin() + i); break; }
    }
  }

  size_t eff_total = 0;
  for (int c : cnt) eff_total += c;
  bin_upper_bound_ = GreedyFindBin(dv, cnt, usable_bins, eff_total, min_data_in_bin);
  if (forced_bounds != nullptr && !forced_bounds->empty()) {
    // forcedbins_filename (reference DatasetLoader ForceBins parity): the forced
    // upper bounds are kept exactly; quantile boundaries fill the remaining
    // budget, dropping the ones with the smallest neighbour gap when over it
    std::vector<std::pair<double, bool>> merged;  // (bound, is_forced)
    for (double b : bin_upper_bound_)
      if (std::isfinite(b)) merged.push_back({b, false});
    for (double b : *forced_bounds)
      if (std::isfinite(b)) merged.push_back({b, true});
    std::sort(merged.begin(), merged.end());
    // dedup: forced wins
    std::vector<std::pair<double, bool>> uniq;
    for (auto& e : merged) {
      if (!uniq.empty() && uniq.back().first == e.first)
        uniq.back().second = uniq.back().second || e.second;
      else
        uniq.push_back(e);
    }
    const int budget = std::max(1, usable_bins - 1);  // +inf terminator re-added below
    while (static_cast<int>(uniq.size()) > budget) {
      int victim = -1;
      double best_gap = std::numeric_limits<double>::infinity();
      for (size_t i = 0; i < uniq.size(); ++i) {
        if (uniq[i].second) continue;  // never drop a forced bound
        const double lo2 = i == 0 ? -std::numeric_limits<double>::infinity()
                                  : uniq[i - 1].first;
        const double hi2 = i + 1 < uniq.size()
                               ? uniq[i + 1].first
                               : std::numeric_limits<double>::infinity();
        const double gap = std::min(uniq[i].first - lo2, hi2 - uniq[i].first);
        if (gap < best_gap) {
          best_gap = gap;
          victim = static_cast<int>(i);
        }
      }
      if (victim < 0) break;  // all forced: honor them even over budget
      uniq.erase(uniq.begin() + victim);
    }
    bin_upper_bound_.clear();
    for (auto& e : uniq) bin_upper_bound_.push_back(e.first);
    bin_upper_bound_.push_back(std::numeric_limits<double>::infinity());
  }
  num_numeric_bin_ = static_cast<int>(bin_upper_bound_.size());
  if (num_numeric_bin_ == 0) {
    bin_upper_bound_.push_back(std::numeric_limits<double>::infinity());
    num_numeric_bin_ = 1;
  }
  num_bin_ = num_numeric_bin_;
  if (missing_type_ == MissingType::kNaN || missing_type_ == MissingType::kZero) num_bin_ += 1;

  is_trivial_ = (num_bin_ <= 1) || (pre_filter && num_bin_ <= 1);
  if (num_numeric_bin_ <= 1 && missing_type_ == MissingType::kNone) is_trivial_ = true;

  // most frequent bin (for reference-compatible metadata; dense storage keeps all bins)
  std::vector<size_t> bin_cnt(num_bin_, 0);
  for (size_t i = 0; i < dv.size(); ++i) {
    uint32_t b = ValueToBin(dv[i]);
    bin_cnt[b] += cnt[i];
  }
  if (missing_type_ == MissingType::kNaN) bin_cnt[num_bin_ - 1] += na_cnt;
  if (missing_type_ == MissingType::kZero) bin_cnt[num_bin_ - 1] += zero_cnt;
  most_freq_bin_ = static_cast<uint32_t>(std::distance(
      bin_cnt.begin(), std::max_element(bin_cnt.begin(), bin_cnt.end())));
  default_bin_ = ValueToBin(0.0);
  sparse_rate_ = total_sample_cnt > 0
      ? static_cast<double>(bin_cnt[default_bin_]) / total_sample_cnt : 0.0;
}

std::string BinMapper::ToFeatureInfoString() const {
  if (is_trivial_) return "none";
  std::stringstream ss;
  if (bin_type_ == BinType::kNumerical) {
    ss << "[" << Common::DoubleToStr(min_val_) << ":" << Common::DoubleToStr(max_val_) << "]";
  } else {
    for (size_t i = 1; i < bin_2_categorical_.size(); ++i) {
      if (i > 1) ss << ":";
      ss << bin_2_categorical_[i];
    }
  }
  return ss.str();
}

std::string BinMapper::ToString() const {
  std::stringstream ss;
  ss << num_bin_ << " " << num_numeric_bin_ << " " << static_cast<int>(is_trivial_) << " "
     << static_cast<int>(bin_type_) << " " << static_cast<int>(missing_type_) << " "
     << most_freq_bin_ << " " << default_bin_ << " " << Common::DoubleToStr(sparse_rate_) << " "
     << Common::DoubleToStr(min_val_) << " " << Common::DoubleToStr(max_val_) << "\n";
  ss << Common::ArrayToString(bin_upper_bound_.data(), bin_upper_bound_.size()) << "\n";
  ss << Common::ArrayToString(bin_2_categorical_.data(), bin_2_categorical_.size()) << "\n";
  return ss.str();
}

void BinMapper::FromString(const std::string& s) {
  auto lines = Common::Split(s.c_str(), '\n');
  MIGBM_CHECK_GE(lines.size(), 3u);
  auto head = Common::SplitAny(lines[0].c_str(), " ");
  MIGBM_CHECK_GE(head.size(), 10u);
  num_bin_ = atoi(head[0].c_str());
  num_numeric_bin_ = atoi(head[1].c_str());
  is_trivial_ = atoi(head[2].c_str()) != 0;
  bin_type_ = static_cast<BinType>(atoi(head[3].c_str()));
  missing_type_ = static_cast<MissingType>(atoi(head[4].c_str()));
  most_freq_bin_ = static_cast<uint32_t>(atoi(head[5].c_str()));
  default_bin_ = static_cast<uint32_t>(atoi(head[6].c_str()));
  sparse_rate_ = Common::Atof(head[7].c_str());
  min_val_ = Common::Atof(head[8].c_str());
  max_val_ = Common::Atof(head[9].c_str());
  Common::StringToArray<double>(lines[1], ' ', &bin_upper_bound_);
  Common::StringToArray<int>(lines[2], ' ', &bin_2_categorical_);
  categorical_2_bin_.clear();
  for (size_t i = 1; i < bin_2_categorical_.size(); ++i)
    categorical_2_bin_[bin_2_categorical_[i]] = static_cast<uint32_t>(i);
}

}  // namespace migbm
