/*! migbm Tree implementation: split recording, model-text v4 serialization, JSON dump,
 *  parsing. Format parity target: reference src/io/tree.cpp:343-500 (field names/order). */
#include "migbm/tree.h"
#include "migbm/dataset.h"

#include <functional>
#include <iomanip>
#include <map>
#include <sstream>

namespace migbm {

Tree::Tree(int max_leaves, bool /*track_branch_features*/, bool is_linear)
    : max_leaves_(max_leaves), num_leaves_(1), is_linear_(is_linear) {
  const int m = max_leaves_;
  left_child_.resize(m - 1);
  right_child_.resize(m - 1);
  split_feature_inner_.resize(m - 1);
  split_feature_.resize(m - 1);
  threshold_in_bin_.resize(m - 1);
  threshold_.resize(m - 1);
  decision_type_.assign(m - 1, 0);
  split_gain_.resize(m - 1);
  internal_value_.assign(m - 1, 0.0);
  internal_weight_.assign(m - 1, 0.0);
  internal_count_.assign(m - 1, 0);
  leaf_value_.assign(m, 0.0);
  leaf_weight_.assign(m, 0.0);
  leaf_count_.assign(m, 0);
  leaf_depth_.assign(m, 0);
  leaf_parent_.assign(m, -1);
  cat_boundaries_.push_back(0);
}

void Tree::RecordSplit(int leaf, int new_node, int feature, int real_feature,
                       double left_value, double right_value, int left_cnt, int right_cnt,
                       double left_weight, double right_weight, float gain) {
  split_feature_inner_[new_node] = feature;
  split_feature_[new_node] = real_feature;
  split_gain_[new_node] = gain;
  // hook up parent
  int parent = leaf_parent_[leaf];
  if (parent >= 0) {
    if (left_child_[parent] == ~leaf) left_child_[parent] = new_node;
    else right_child_[parent] = new_node;
  }
  // internal stats = combination of children
  internal_value_[new_node] = (left_weight * left_value + right_weight * right_value) /
                              std::max(left_weight + right_weight, kEpsilon);
  internal_weight_[new_node] = left_weight + right_weight;
  internal_count_[new_node] = left_cnt + right_cnt;
  left_child_[new_node] = ~leaf;
  right_child_[new_node] = ~num_leaves_;
  leaf_parent_[leaf] = new_node;
  leaf_parent_[num_leaves_] = new_node;
  leaf_value_[leaf] = std::isnan(left_value) ? 0.0 : left_value;
  leaf_value_[num_leaves_] = std::isnan(right_value) ? 0.0 : right_value;
  leaf_weight_[leaf] = left_weight;
  leaf_weight_[num_leaves_] = right_weight;
  leaf_count_[leaf] = left_cnt;
  leaf_count_[num_leaves_] = right_cnt;
  leaf_depth_[num_leaves_] = leaf_depth_[leaf] + 1;
  leaf_depth_[leaf] += 1;
}

int Tree::Split(int leaf, int feature, int real_feature, uint32_t threshold_bin,
                double threshold_double, double left_value, double right_value, int left_cnt,
                int right_cnt, double left_weight, double right_weight, float gain,
                MissingType missing_type, bool default_left) {
  const int new_node = num_leaves_ - 1;
  decision_type_[new_node] = 0;
  if (default_left) decision_type_[new_node] |= kDefaultLeftMask;
  decision_type_[new_node] |= static_cast<int8_t>(static_cast<int>(missing_type) << 2);
  threshold_in_bin_[new_node] = threshold_bin;
  threshold_[new_node] = threshold_double;
  RecordSplit(leaf, new_node, feature, real_feature, left_value, right_value, left_cnt,
              right_cnt, left_weight, right_weight, gain);
  ++num_leaves_;
  return num_leaves_ - 1;
}

int Tree::SplitCategorical(int leaf, int feature, int real_feature,
                           const uint32_t* threshold_bitset, int n_words, double left_value,
                           double right_value, int left_cnt, int right_cnt, double left_weight,
                           double right_weight, float gain, MissingType missing_type) {
  const int new_node = num_leaves_ - 1;
  decision_type_[new_node] = kCategoricalMask;
  decision_type_[new_node] |= static_cast<int8_t>(static_cast<int>(missing_type) << 2);
  threshold_in_bin_[new_node] = static_cast<uint32_t>(num_cat_);
  threshold_[new_node] = static_cast<double>(num_cat_);
  cat_boundaries_.push_back(cat_boundaries_.back() + n_words);
  for (int i = 0; i < n_words; ++i) cat_threshold_.push_back(threshold_bitset[i]);
  ++num_cat_;
  RecordSplit(leaf, new_node, feature, real_feature, left_value, right_value, left_cnt,
              right_cnt, left_weight, right_weight, gain);
  ++num_leaves_;
  return num_leaves_ - 1;
}

void Tree::OverrideLeafCounts(const std::vector<int>& counts) {
  for (int l = 0; l < num_leaves_ && l < static_cast<int>(counts.size()); ++l)
    leaf_count_[l] = counts[l];
  if (num_leaves_ <= 1) return;
  std::function<int(int)> rec = [&](int node) -> int {
    if (node < 0) return leaf_count_[~node];
    int c = rec(left_child_[node]) + rec(right_child_[node]);
    internal_count_[node] = c;
    return c;
  };
  rec(0);
}

double Tree::GetUpperBoundValue() const {
  double mx = leaf_value_[0];
  for (int i = 1; i < num_leaves_; ++i) mx = std::max(mx, leaf_value_[i]);
  return mx;
}
double Tree::GetLowerBoundValue() const {
  double mn = leaf_value_[0];
  for (int i = 1; i < num_leaves_; ++i) mn = std::min(mn, leaf_value_[i]);
  return mn;
}

std::string Tree::ToString() const {
  std::stringstream ss;
  ss.precision(17);
  const int ni = num_leaves_ - 1;
  ss << "num_leaves=" << num_leaves_ << '\n';
  ss << "num_cat=" << num_cat_ << '\n';
  ss << "split_feature=" << Common::ArrayToString(split_feature_.data(), ni) << '\n';
  ss << "split_gain=" << Common::ArrayToString(split_gain_.data(), ni) << '\n';
  ss << "threshold=" << Common::ArrayToString(threshold_.data(), ni) << '\n';
  {
    // decision_type serialized as ints
    std::vector<int> dt(ni);
    for (int i = 0; i < ni; ++i) dt[i] = decision_type_[i];
    ss << "decision_type=" << Common::ArrayToString(dt.data(), ni) << '\n';
  }
  ss << "left_child=" << Common::ArrayToString(left_child_.data(), ni) << '\n';
  ss << "right_child=" << Common::ArrayToString(right_child_.data(), ni) << '\n';
  ss << "leaf_value=" << Common::ArrayToString(leaf_value_.data(), num_leaves_) << '\n';
  ss << "leaf_weight=" << Common::ArrayToString(leaf_weight_.data(), num_leaves_) << '\n';
  ss << "leaf_count=" << Common::ArrayToString(leaf_count_.data(), num_leaves_) << '\n';
  ss << "internal_value=" << Common::ArrayToString(internal_value_.data(), ni) << '\n';
  ss << "internal_weight=" << Common::ArrayToString(internal_weight_.data(), ni) << '\n';
  ss << "internal_count=" << Common::ArrayToString(internal_count_.data(), ni) << '\n';
  if (num_cat_ > 0) {
    ss << "cat_boundaries=" << Common::ArrayToString(cat_boundaries_.data(), cat_boundaries_.size()) << '\n';
    ss << "cat_threshold=" << Common::ArrayToString(cat_threshold_.data(), cat_threshold_.size()) << '\n';
  }
  ss << "is_linear=" << (is_linear_ ? 1 : 0) << '\n';
  if (is_linear_) {
    std::vector<double> consts(num_leaves_);
    std::vector<int> nfeat(num_leaves_);
    std::vector<int> flat_feats;
    std::vector<double> flat_coefs;
    for (int l = 0; l < num_leaves_; ++l) {
      consts[l] = l < static_cast<int>(leaf_const_.size()) ? leaf_const_[l] : 0.0;
      const auto& ff = l < static_cast<int>(leaf_features_.size()) ? leaf_features_[l]
                                                                   : std::vector<int>();
      nfeat[l] = static_cast<int>(ff.size());
      for (int f : ff) flat_feats.push_back(f);
      if (l < static_cast<int>(leaf_coeff_.size()))
        for (double c : leaf_coeff_[l]) flat_coefs.push_back(c);
    }
    ss << "leaf_const=" << Common::ArrayToString(consts.data(), consts.size()) << '\n';
    ss << "num_features=" << Common::ArrayToString(nfeat.data(), nfeat.size()) << '\n';
    ss << "leaf_features=" << Common::ArrayToString(flat_feats.data(), flat_feats.size()) << '\n';
    ss << "leaf_coeff=" << Common::ArrayToString(flat_coefs.data(), flat_coefs.size()) << '\n';
  }
  ss << "shrinkage=" << Common::DoubleToStr(shrinkage_) << '\n';
  return ss.str();
}

namespace {
std::map<std::string, std::string> ParseKV(const char* str, size_t* used_len) {
  std::map<std::string, std::string> kv;
  const char* p = str;
  size_t consumed = 0;
  while (*p) {
    const char* eol = strchr(p, '\n');
    size_t len = eol ? static_cast<size_t>(eol - p) : strlen(p);
    std::string line(p, len);
    line = Common::Trim(line);
    if (line.empty()) {
      consumed = (eol ? (eol - str) + 1 : strlen(str));
      if (!kv.empty()) break;  // blank line ends the tree block (after content started)
      p = eol ? eol + 1 : p + len;
      continue;
    }
    if (Common::StartsWith(line, "Tree=")) {
      if (!kv.empty()) break;
      p = eol ? eol + 1 : p + len;
      consumed = p - str;
      continue;
    }
    if (Common::StartsWith(line, "end of trees")) break;
    auto eq = line.find('=');
    if (eq != std::string::npos) kv[line.substr(0, eq)] = line.substr(eq + 1);
    p = eol ? eol + 1 : p + len;
    consumed = p - str;
    if (!eol) break;
  }
  if (used_len) *used_len = consumed;
  return kv;
}
}  // namespace

Tree::Tree(const char* str, size_t* used_len) {
  bin_thresholds_valid_ = false;  // model text carries only real-valued thresholds
  auto kv = ParseKV(str, used_len);
  auto get = [&](const char* k) -> const std::string& {
    static const std::string empty;
    auto it = kv.find(k);
    return it == kv.end() ? empty : it->second;
  };
  num_leaves_ = atoi(get("num_leaves").c_str());
  num_cat_ = atoi(get("num_cat").c_str());
  is_linear_ = atoi(get("is_linear").c_str()) != 0;
  shrinkage_ = Common::Atof(get("shrinkage").c_str());
  if (shrinkage_ == 0.0) shrinkage_ = 1.0;
  max_leaves_ = std::max(num_leaves_, 1);
  leaf_value_.assign(max_leaves_, 0.0);
  Common::StringToArray<double>(get("leaf_value"), ' ', &leaf_value_);
  if (num_leaves_ <= 1) { cat_boundaries_.push_back(0); return; }
  Common::StringToArray<int>(get("split_feature"), ' ', &split_feature_);
  split_feature_inner_ = split_feature_;  // standalone model: inner == real
  Common::StringToArray<float>(get("split_gain"), ' ', &split_gain_);
  Common::StringToArray<double>(get("threshold"), ' ', &threshold_);
  {
    std::vector<int> dt;
    Common::StringToArray<int>(get("decision_type"), ' ', &dt);
    decision_type_.resize(dt.size());
    for (size_t i = 0; i < dt.size(); ++i) decision_type_[i] = static_cast<int8_t>(dt[i]);
  }
  Common::StringToArray<int>(get("left_child"), ' ', &left_child_);
  Common::StringToArray<int>(get("right_child"), ' ', &right_child_);
  Common::StringToArray<double>(get("leaf_weight"), ' ', &leaf_weight_);
  Common::StringToArray<int>(get("leaf_count"), ' ', &leaf_count_);
  Common::StringToArray<double>(get("internal_value"), ' ', &internal_value_);
  Common::StringToArray<double>(get("internal_weight"), ' ', &internal_weight_);
  Common::StringToArray<int>(get("internal_count"), ' ', &internal_count_);
  if (num_cat_ > 0) {
    Common::StringToArray<int>(get("cat_boundaries"), ' ', &cat_boundaries_);
    Common::StringToArray<uint32_t>(get("cat_threshold"), ' ', &cat_threshold_);
  } else {
    cat_boundaries_.push_back(0);
  }
  if (is_linear_) {
    SetLinear(true);
    std::vector<double> consts;
    std::vector<int> nfeat, flat_feats;
    std::vector<double> flat_coefs;
    Common::StringToArray<double>(get("leaf_const"), ' ', &consts);
    Common::StringToArray<int>(get("num_features"), ' ', &nfeat);
    Common::StringToArray<int>(get("leaf_features"), ' ', &flat_feats);
    Common::StringToArray<double>(get("leaf_coeff"), ' ', &flat_coefs);
    size_t off = 0;
    for (int l = 0; l < num_leaves_ && l < static_cast<int>(nfeat.size()); ++l) {
      leaf_const_[l] = l < static_cast<int>(consts.size()) ? consts[l] : 0.0;
      const int k = nfeat[l];
      std::vector<int> ff(flat_feats.begin() + off, flat_feats.begin() + off + k);
      std::vector<double> cc(flat_coefs.begin() + off, flat_coefs.begin() + off + k);
      leaf_features_[l] = ff;
      leaf_features_inner_[l] = ff;
      leaf_coeff_[l] = cc;
      off += k;
    }
  }
  threshold_in_bin_.assign(num_leaves_ - 1, 0);
  leaf_depth_.assign(num_leaves_, 0);
  leaf_parent_.assign(num_leaves_, -1);
  leaf_weight_.resize(num_leaves_, 0.0);
  leaf_count_.resize(num_leaves_, 0);
  internal_value_.resize(num_leaves_ - 1, 0.0);
  internal_weight_.resize(num_leaves_ - 1, 0.0);
  internal_count_.resize(num_leaves_ - 1, 0);
  split_gain_.resize(num_leaves_ - 1, 0.0f);
}

std::string Tree::ToJSON() const {
  std::stringstream ss;
  ss.precision(17);
  ss << "{";
  ss << "\"num_leaves\":" << num_leaves_ << ",";
  ss << "\"num_cat\":" << num_cat_ << ",";
  ss << "\"shrinkage\":" << shrinkage_ << ",";
  ss << "\"tree_structure\":";
  // recursive node dump
  std::function<void(int)> dump = [&](int node) {
    if (node >= 0) {
      ss << "{\"split_index\":" << node
         << ",\"split_feature\":" << split_feature_[node]
         << ",\"split_gain\":" << split_gain_[node];
      if (IsCategoricalSplit(node)) {
        // reference JSON: categorical threshold = "c1||c2||..." category values
        const int cat_idx = static_cast<int>(threshold_[node]);
        const uint32_t* bits = cat_threshold_.data() + cat_boundaries_[cat_idx];
        const int n_words = cat_boundaries_[cat_idx + 1] - cat_boundaries_[cat_idx];
        std::string cats;
        for (int w = 0; w < n_words; ++w) {
          for (int b = 0; b < 32; ++b) {
            if ((bits[w] >> b) & 1u) {
              if (!cats.empty()) cats += "||";
              cats += std::to_string(w * 32 + b);
            }
          }
        }
        ss << ",\"threshold\":\"" << cats << "\"";
      } else {
        ss << ",\"threshold\":" << Common::DoubleToStr(threshold_[node]);
      }
      ss << ",\"decision_type\":\"" << (IsCategoricalSplit(node) ? "==" : "<=") << "\""
         << ",\"default_left\":" << ((decision_type_[node] & kDefaultLeftMask) ? "true" : "false")
         << ",\"missing_type\":\"";
      int mt = (decision_type_[node] >> 2) & 3;
      ss << (mt == 0 ? "None" : (mt == 1 ? "Zero" : "NaN")) << "\""
         << ",\"internal_value\":" << internal_value_[node]
         << ",\"internal_weight\":" << internal_weight_[node]
         << ",\"internal_count\":" << internal_count_[node]
         << ",\"left_child\":";
      dump(left_child_[node]);
      ss << ",\"right_child\":";
      dump(right_child_[node]);
      ss << "}";
    } else {
      int leaf = ~node;
      ss << "{\"leaf_index\":" << leaf
         << ",\"leaf_value\":" << Common::DoubleToStr(leaf_value_[leaf])
         << ",\"leaf_weight\":" << leaf_weight_[leaf]
         << ",\"leaf_count\":" << leaf_count_[leaf] << "}";
    }
  };
  if (num_leaves_ > 1) dump(0);
  else dump(~0);
  ss << "}";
  return ss.str();
}



std::string Tree::ToIfElse(int index) const {
  // Standalone nested-if codegen. Decision semantics mirror NumericalDecision /
  // CategoricalDecision exactly (missing handling included); the generated file
  // depends only on <cmath>.
  std::stringstream ss;
  ss << std::setprecision(17);
  // categorical bitset words for this tree, one flat static array
  if (!cat_threshold_.empty()) {
    ss << "static const unsigned int cat_bits_" << index << "[] = {";
    for (size_t i = 0; i < cat_threshold_.size(); ++i) {
      if (i) ss << ",";
      ss << cat_threshold_[i] << "u";
    }
    ss << "};\n";
  }
  // recursive emitters for value and leaf-index variants
  std::function<void(int, int, bool)> emit = [&](int node, int depth, bool leaf_index) {
    std::string ind(static_cast<size_t>(depth) * 2 + 2, ' ');
    if (node < 0) {  // leaf
      int leaf = ~node;
      if (leaf_index) {
        ss << ind << "return " << leaf << ";\n";
        return;
      }
      if (is_linear_ && leaf < static_cast<int>(leaf_coeff_.size()) &&
          !leaf_coeff_[leaf].empty()) {
        // linear leaf: const + Σ coeff·x, falling back to the piecewise-constant
        // value when any used feature is missing (LeafOutputLinear semantics)
        ss << ind << "{\n";
        std::string cond;
        for (size_t i = 0; i < leaf_coeff_[leaf].size(); ++i) {
          if (i) cond += " || ";
          cond += "std::isnan(arr[" + std::to_string(leaf_features_[leaf][i]) + "])";
        }
        ss << ind << "  if (" << cond << ") return " << leaf_value_[leaf] << ";\n";
        ss << ind << "  return " << leaf_const_[leaf];
        for (size_t i = 0; i < leaf_coeff_[leaf].size(); ++i)
          ss << " + " << leaf_coeff_[leaf][i] << " * arr["
             << leaf_features_[leaf][i] << "]";
        ss << ";\n" << ind << "}\n";
        return;
      }
      ss << ind << "return " << leaf_value_[leaf] << ";\n";
      return;
    }
    const int fid = split_feature_[node];
    if (decision_type_[node] & kCategoricalMask) {
      const int cat_idx = static_cast<int>(threshold_[node]);
      const int off = cat_boundaries_[cat_idx];
      const int n_words = cat_boundaries_[cat_idx + 1] - off;
      ss << ind << "if (CategoricalDecision(arr[" << fid << "], cat_bits_" << index
         << " + " << off << ", " << n_words << ")) {\n";
    } else {
      const int8_t dt = decision_type_[node];
      const int missing_type = (dt >> 2) & 3;
      const bool default_left = (dt & kDefaultLeftMask) != 0;
      ss << ind << "if (NumericalDecision(arr[" << fid << "], " << missing_type << ", "
         << (default_left ? "true" : "false") << ", " << threshold_[node] << ")) {\n";
    }
    emit(left_child_[node], depth + 1, leaf_index);
    ss << ind << "} else {\n";
    emit(right_child_[node], depth + 1, leaf_index);
    ss << ind << "}\n";
  };
  ss << "double PredictTree" << index << "(const double* arr) {\n";
  if (num_leaves_ <= 1) {
    ss << "  return " << (leaf_value_.empty() ? 0.0 : leaf_value_[0]) << ";\n";
  } else {
    emit(0, 0, false);
  }
  ss << "}\n\n";
  ss << "int PredictTree" << index << "LeafIndex(const double* arr) {\n";
  if (num_leaves_ <= 1) {
    ss << "  return 0;\n";
  } else {
    emit(0, 0, true);
  }
  ss << "}\n\n";
  return ss.str();
}

}  // namespace migbm
