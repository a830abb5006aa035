/*!
 * migbm Tree — flat-array decision tree model.
 * Capability parity target: reference include/LightGBM/tree.h + src/io/tree.cpp
 * (Split/SplitCategorical, Predict, ToString model-text-v4 block, ToJSON).
 * Fresh implementation; the text serialization field names match the reference
 * format so models interchange.
 */
#ifndef MIGBM_TREE_H_
#define MIGBM_TREE_H_

#include "common.h"
#include "bin.h"

#include <string>
#include <vector>

namespace migbm {

class Tree {
 public:
  // decision_type bit layout (must match reference for model-text compat)
  static constexpr int8_t kCategoricalMask = 1;
  static constexpr int8_t kDefaultLeftMask = 2;

  explicit Tree(int max_leaves, bool track_branch_features = false, bool is_linear = false);
  explicit Tree(const char* str, size_t* used_len);  // parse from model text

  /*! Split leaf -> (left=new internal node's left child keeps `leaf` id, right=new leaf).
   *  Returns index of the new right leaf (== num_leaves_-1 before increment semantics of ref:
   *  right child leaf id = num_leaves_). */
  int Split(int leaf, int feature, int real_feature, uint32_t threshold_bin,
            double threshold_double, double left_value, double right_value,
            int left_cnt, int right_cnt, double left_weight, double right_weight,
            float gain, MissingType missing_type, bool default_left);

  /*! Categorical split: threshold is a bitset over bin ids (in-bitset -> left). */
  int SplitCategorical(int leaf, int feature, int real_feature, const uint32_t* threshold_bitset,
                       int n_words, double left_value, double right_value, int left_cnt,
                       int right_cnt, double left_weight, double right_weight, float gain,
                       MissingType missing_type);

  double Predict(const double* feature_values) const {
    if (num_leaves_ > 1) {
      int node = 0;
      while (node >= 0) node = Decision(feature_values, node);
      return is_linear_ ? LeafOutputLinear(~node, feature_values) : LeafOutput(~node);
    }
    return num_leaves_ == 1 ? leaf_value_[0] : 0.0;
  }
  int PredictLeafIndex(const double* feature_values) const {
    if (num_leaves_ > 1) {
      int node = 0;
      while (node >= 0) node = Decision(feature_values, node);
      return ~node;
    }
    return 0;
  }

  int num_leaves() const { return num_leaves_; }
  double LeafOutput(int leaf) const { return leaf_value_[leaf]; }
  void SetLeafOutput(int leaf, double v) { leaf_value_[leaf] = v; }
  int leaf_count(int leaf) const { return leaf_count_[leaf]; }
  double leaf_weight(int leaf) const { return leaf_weight_[leaf]; }
  int split_feature(int node) const { return split_feature_[node]; }
  int split_feature_inner(int node) const { return split_feature_inner_[node]; }
  double threshold(int node) const { return threshold_[node]; }
  uint32_t threshold_in_bin(int node) const { return threshold_in_bin_[node]; }
  float split_gain(int node) const { return split_gain_[node]; }
  int left_child(int node) const { return left_child_[node]; }
  int right_child(int node) const { return right_child_[node]; }
  int8_t decision_type(int node) const { return decision_type_[node]; }
  bool IsCategoricalSplit(int node) const { return (decision_type_[node] & kCategoricalMask) != 0; }
  double shrinkage() const { return shrinkage_; }
  int leaf_depth(int leaf) const { return leaf_depth_[leaf]; }
  int leaf_parent(int leaf) const { return leaf_parent_[leaf]; }
  double internal_value(int node) const { return internal_value_[node]; }
  double InternalCountSafe(int node) const {
    return node < static_cast<int>(internal_count_.size()) ? internal_count_[node] : 1.0;
  }

  void Shrinkage(double rate) {
    shrinkage_ *= rate;
    for (int i = 0; i < num_leaves_; ++i) leaf_value_[i] = MaybeRound(leaf_value_[i] * rate);
    for (int i = 0; i < num_leaves_ - 1; ++i) internal_value_[i] *= rate;
    if (is_linear_) {
      for (int i = 0; i < num_leaves_; ++i) {
        leaf_const_[i] *= rate;
        for (auto& c : leaf_coeff_[i]) c *= rate;
      }
    }
  }
  void AddBias(double val) {
    for (int i = 0; i < num_leaves_; ++i) leaf_value_[i] = MaybeRound(val + leaf_value_[i]);
    for (int i = 0; i < num_leaves_ - 1; ++i) internal_value_[i] += val;
    if (is_linear_) {
      for (int i = 0; i < num_leaves_; ++i) leaf_const_[i] += val;
    }
  }
  void AsConstantTree(double val, int count = 0) {
    num_leaves_ = 1;
    shrinkage_ = 1.0;
    leaf_value_[0] = val;
    leaf_count_[0] = count;
  }

  /*! score[i] += tree(data row i) over an index set (used by score updater on CPU). */
  void AddPredictionToScore(const class Dataset* data, data_size_t num_data, double* score) const;
  /*! bin->representative-value walk for trees WITHOUT bin thresholds (loaded models). */
  void AddPredictionToScoreByValue(const class Dataset* data, data_size_t num_data,
                                   double* score) const;
  void AddPredictionToScore(const class Dataset* data, const data_size_t* used_indices,
                            data_size_t num_data, double* score) const;

  /*! replace leaf counts with exact values and recompute internal counts bottom-up
   *  (the HIP learner builds with hessian-approx counts and fixes them here). */
  void OverrideLeafCounts(const std::vector<int>& counts);

  std::string ToString() const;   // model-text v4 tree block
  /*! standalone C++ if-else code for this tree (reference parity: convert_model
   *  with convert_model_language=cpp). Emits PredictTree<idx> and
   *  PredictTree<idx>LeafIndex functions; categorical bitsets become static arrays. */
  std::string ToIfElse(int index) const;
  std::string ToJSON() const;

  /*! leaf ids in pre-order; maps categorical bitset words. */
  const std::vector<int>& cat_boundaries() const { return cat_boundaries_; }
  const std::vector<uint32_t>& cat_threshold() const { return cat_threshold_; }
  int num_cat() const { return num_cat_; }
  bool is_linear() const { return is_linear_; }
  void SetLinear(bool v) {
    is_linear_ = v;
    if (v && leaf_const_.empty()) {
      leaf_const_.assign(max_leaves_, 0.0);
      leaf_features_.resize(max_leaves_);
      leaf_features_inner_.resize(max_leaves_);
      leaf_coeff_.resize(max_leaves_);
    }
  }
  /*! set a leaf's linear model: output = const + sum(coeff[i] * x[feat[i]]) */
  void SetLeafLinear(int leaf, double constant, const std::vector<int>& feats_real,
                     const std::vector<int>& feats_inner, const std::vector<double>& coeff) {
    leaf_const_[leaf] = constant;
    leaf_features_[leaf] = feats_real;
    leaf_features_inner_[leaf] = feats_inner;
    leaf_coeff_[leaf] = coeff;
  }
  const std::vector<int>& leaf_features_inner(int leaf) const {
    return leaf_features_inner_[leaf];
  }
  const std::vector<double>& leaf_coeffs(int leaf) const { return leaf_coeff_[leaf]; }
  double leaf_const(int leaf) const { return leaf_const_[leaf]; }
  double LeafOutputLinear(int leaf, const double* feature_values) const {
    if (!is_linear_ || leaf_coeff_[leaf].empty()) return leaf_value_[leaf];
    double out = leaf_const_[leaf];
    for (size_t i = 0; i < leaf_coeff_[leaf].size(); ++i) {
      const double v = feature_values[leaf_features_[leaf][i]];
      if (std::isnan(v)) return leaf_value_[leaf];  // missing -> piecewise-constant value
      out += leaf_coeff_[leaf][i] * v;
    }
    return out;
  }

  /*! expected maximum value |leaf output| for bound calc */
  double GetUpperBoundValue() const;
  double GetLowerBoundValue() const;

  inline int Decision(const double* values, int node) const {
    if (IsCategoricalSplit(node)) return CategoricalDecision(values[split_feature_[node]], node);
    return NumericalDecision(values[split_feature_[node]], node);
  }

  inline int NumericalDecision(double value, int node) const {
    const int8_t dt = decision_type_[node];
    const uint8_t missing_type = (dt >> 2) & 3;
    if (std::isnan(value) && missing_type != 2) value = 0.0;
    if ((missing_type == 1 && value == 0.0) || (missing_type == 2 && std::isnan(value))) {
      return (dt & kDefaultLeftMask) ? left_child_[node] : right_child_[node];
    }
    return value <= threshold_[node] ? left_child_[node] : right_child_[node];
  }

  inline int CategoricalDecision(double value, int node) const {
    int cat = static_cast<int>(value);
    if (std::isnan(value) || cat < 0) return right_child_[node];
    const int cat_idx = static_cast<int>(threshold_[node]);
    const uint32_t* bits = cat_threshold_.data() + cat_boundaries_[cat_idx];
    const int n_words = cat_boundaries_[cat_idx + 1] - cat_boundaries_[cat_idx];
    if ((cat >> 5) < n_words && ((bits[cat >> 5] >> (cat & 31)) & 1)) return left_child_[node];
    return right_child_[node];
  }

 private:
  double MaybeRound(double v) const { return v; }
  void RecordSplit(int leaf, int new_node, int feature, int real_feature, double left_value,
                   double right_value, int left_cnt, int right_cnt, double left_weight,
                   double right_weight, float gain);

  int max_leaves_;
  int num_leaves_;
  int num_cat_ = 0;
  bool is_linear_ = false;
  // false for text-loaded trees: threshold_in_bin_ / bin-space cat bitsets are
  // absent, so dataset walks must route by real values (ByValue)
  bool bin_thresholds_valid_ = true;
  double shrinkage_ = 1.0;
  // per internal node (num_leaves_-1 entries)
  std::vector<int> left_child_, right_child_;
  std::vector<int> split_feature_inner_, split_feature_;
  std::vector<uint32_t> threshold_in_bin_;
  std::vector<double> threshold_;
  std::vector<int8_t> decision_type_;
  std::vector<float> split_gain_;
  std::vector<double> internal_value_, internal_weight_;
  std::vector<int> internal_count_;
  // per leaf
  std::vector<double> leaf_value_, leaf_weight_;
  std::vector<int> leaf_count_, leaf_depth_, leaf_parent_;
  // categorical bitsets
  std::vector<int> cat_boundaries_;
  std::vector<uint32_t> cat_threshold_;
  // linear leaves (linear_tree=true)
  std::vector<double> leaf_const_;
  std::vector<std::vector<int>> leaf_features_, leaf_features_inner_;
  std::vector<std::vector<double>> leaf_coeff_;
};

}  // namespace migbm

#endif  // MIGBM_TREE_H_
