/*! migbm model-text v4 writer/parser + JSON dump + feature importance.
 *  Format parity target: reference src/boosting/gbdt_model_text.cpp:314-405 — field
 *  names and layout match so models interchange with the reference ecosystem. */
#include "migbm/boosting.h"

#include <iomanip>
#include <map>
#include <sstream>

namespace migbm {

std::vector<double> GBDT::FeatureImportance(int num_iter, int importance_type) const {
  int total = num_iter <= 0 ? static_cast<int>(models_.size())
                            : std::min<int>(static_cast<int>(models_.size()),
                                            num_iter * num_tree_per_iteration_);
  std::vector<double> imp(max_feature_idx_ + 1, 0.0);
  for (int t = 0; t < total; ++t) {
    const Tree* tree = models_[t].get();
    for (int n = 0; n < tree->num_leaves() - 1; ++n) {
      if (importance_type == 0) imp[tree->split_feature(n)] += 1.0;
      else imp[tree->split_feature(n)] += tree->split_gain(n);
    }
  }
  return imp;
}

std::string GBDT::SaveModelToString(int start_iter, int num_iter,
                                    int feature_importance_type) const {
  std::stringstream ss;
  ss.precision(17);
  const int total_iters = num_tree_per_iteration_ > 0
                              ? static_cast<int>(models_.size()) / num_tree_per_iteration_ : 0;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  const int start_model = start_iter * num_tree_per_iteration_;
  const int end_model = end_iter * num_tree_per_iteration_;

  ss << SubModelName() << '\n';
  ss << "version=v4" << '\n';
  ss << "num_class=" << num_class_ << '\n';
  ss << "num_tree_per_iteration=" << num_tree_per_iteration_ << '\n';
  ss << "label_index=" << label_idx_ << '\n';
  ss << "max_feature_idx=" << max_feature_idx_ << '\n';
  if (!objective_tostring_.empty()) ss << "objective=" << objective_tostring_ << '\n';
  else if (objective_ != nullptr) ss << "objective=" << objective_->ToString() << '\n';
  else if (!objective_name_.empty()) ss << "objective=" << objective_name_ << '\n';
  if (average_output_) ss << "average_output" << '\n';
  ss << "feature_names=" << Common::Join(feature_names_, " ") << '\n';
  if (!feature_infos_.empty()) ss << "feature_infos=" << Common::Join(feature_infos_, " ") << '\n';
  else {
    std::vector<std::string> none(max_feature_idx_ + 1, "none");
    ss << "feature_infos=" << Common::Join(none, " ") << '\n';
  }

  // tree blocks (sizes first, like the reference, so loaders can pre-split)
  std::vector<std::string> tree_strs;
  std::vector<size_t> tree_sizes;
  for (int t = start_model; t < end_model && t < static_cast<int>(models_.size()); ++t) {
    std::string s = "Tree=" + std::to_string(t - start_model) + "\n" + models_[t]->ToString() + "\n";
    tree_sizes.push_back(s.size());
    tree_strs.push_back(std::move(s));
  }
  ss << "tree_sizes=" << Common::Join(tree_sizes, " ") << '\n';
  ss << '\n';
  for (auto& blk : tree_strs) ss << blk;  // sizes index into this byte stream exactly
  ss << "end of trees" << '\n';
  ss << '\n';

  // feature importances (descending)
  auto imp = FeatureImportance(num_iter, feature_importance_type);
  std::vector<std::pair<double, int>> order;
  for (int i = 0; i <= max_feature_idx_; ++i)
    if (imp[i] > 0) order.emplace_back(imp[i], i);
  std::stable_sort(order.begin(), order.end(),
                   [](auto& a, auto& b) { return a.first > b.first; });
  ss << "feature_importances:" << '\n';
  for (auto& kv : order) {
    std::string name = kv.second < static_cast<int>(feature_names_.size())
                           ? feature_names_[kv.second]
                           : "Column_" + std::to_string(kv.second);
    if (feature_importance_type == 0)
      ss << name << "=" << static_cast<int64_t>(kv.first) << '\n';
    else
      ss << name << "=" << Common::DoubleToStr(kv.first) << '\n';
  }
  ss << '\n';
  ss << "parameters:" << '\n';
  if (!loaded_parameter_.empty()) {
    ss << loaded_parameter_;
  } else if (config_ != nullptr) {
    std::string p = config_->SaveHyperParameters();
    // SaveHyperParameters includes its own "parameters:" header and footer; strip header
    auto pos = p.find('\n');
    ss << p.substr(pos + 1);
  } else {
    ss << "end of parameters" << '\n';
  }
  ss << '\n';
  return ss.str();
}

bool GBDT::SaveModelToFile(int start_iter, int num_iter, int feature_importance_type,
                           const char* filename) const {
  FILE* fp = fopen(filename, "w");
  if (!fp) return false;
  std::string s = SaveModelToString(start_iter, num_iter, feature_importance_type);
  fwrite(s.data(), 1, s.size(), fp);
  fclose(fp);
  return true;
}

bool GBDT::LoadModelFromString(const char* str, size_t len) {
  models_.clear();
  std::string content(str, len);
  auto get_line_val = [&](const char* key) -> std::string {
    std::string k = std::string(key) + "=";
    size_t pos = content.find(k);
    while (pos != std::string::npos && pos != 0 && content[pos - 1] != '\n')
      pos = content.find(k, pos + 1);
    if (pos == std::string::npos) return "";
    size_t eol = content.find('\n', pos);
    return content.substr(pos + k.size(), eol - pos - k.size());
  };
  std::string v;
  v = get_line_val("num_class");
  num_class_ = v.empty() ? 1 : atoi(v.c_str());
  v = get_line_val("num_tree_per_iteration");
  num_tree_per_iteration_ = v.empty() ? num_class_ : atoi(v.c_str());
  v = get_line_val("label_index");
  label_idx_ = v.empty() ? 0 : atoi(v.c_str());
  v = get_line_val("max_feature_idx");
  max_feature_idx_ = v.empty() ? 0 : atoi(v.c_str());
  v = get_line_val("objective");
  if (!v.empty()) {
    objective_tostring_ = v;
    loaded_objective_.reset(ObjectiveFunction::CreateFromModelString(v));
    objective_ = loaded_objective_.get();
    objective_name_ = objective_ ? objective_->GetName() : v;
  }
  average_output_ = content.find("\naverage_output\n") != std::string::npos;
  v = get_line_val("feature_names");
  if (!v.empty()) feature_names_ = Common::SplitAny(v.c_str(), " ");
  v = get_line_val("feature_infos");
  if (!v.empty()) feature_infos_ = Common::SplitAny(v.c_str(), " ");
  // parameters echo
  size_t pstart = content.find("parameters:");
  if (pstart != std::string::npos) {
    size_t pend = content.find("end of parameters", pstart);
    if (pend != std::string::npos)
      loaded_parameter_ = content.substr(pstart + 12, pend - pstart - 12) + "end of parameters\n";
  }
  // trees
  size_t pos = 0;
  while ((pos = content.find("Tree=", pos)) != std::string::npos) {
    if (pos != 0 && content[pos - 1] != '\n') { pos += 5; continue; }
    size_t eol = content.find('\n', pos);
    size_t used = 0;
    models_.emplace_back(new Tree(content.c_str() + eol + 1, &used));
    pos = eol + 1 + (used > 0 ? used : 1);
  }
  iter_ = num_tree_per_iteration_ > 0
              ? static_cast<int>(models_.size()) / num_tree_per_iteration_ : 0;
  Log::Info("Loaded model with %d trees", static_cast<int>(models_.size()));
  return true;
}

std::string GBDT::DumpModel(int start_iter, int num_iter, int feature_importance_type) const {
  std::stringstream ss;
  ss.precision(17);
  const int total_iters = num_tree_per_iteration_ > 0
                              ? static_cast<int>(models_.size()) / num_tree_per_iteration_ : 0;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  ss << "{";
  ss << "\"name\":\"" << SubModelName() << "\",";
  ss << "\"version\":\"v4\",";
  ss << "\"num_class\":" << num_class_ << ",";
  ss << "\"num_tree_per_iteration\":" << num_tree_per_iteration_ << ",";
  ss << "\"label_index\":" << label_idx_ << ",";
  ss << "\"max_feature_idx\":" << max_feature_idx_ << ",";
  if (objective_ != nullptr) ss << "\"objective\":\"" << objective_->ToString() << "\",";
  ss << "\"average_output\":" << (average_output_ ? "true" : "false") << ",";
  ss << "\"feature_names\":[";
  for (size_t i = 0; i < feature_names_.size(); ++i) {
    if (i) ss << ",";
    ss << "\"" << feature_names_[i] << "\"";
  }
  ss << "],";
  ss << "\"monotone_constraints\":[";
  if (config_ != nullptr) {
    for (size_t i = 0; i < config_->monotone_constraints.size(); ++i) {
      if (i) ss << ",";
      ss << config_->monotone_constraints[i];
    }
  }
  ss << "],";
  // feature_infos: {"name": {"min_value":..,"max_value":..,"values":[..]}}
  // parsed back from the "[lo:hi]" / "cat1:cat2:..." strings of the text format
  ss << "\"feature_infos\":{";
  bool first_fi = true;
  for (size_t i = 0; i < feature_infos_.size() && i < feature_names_.size(); ++i) {
    const std::string& fi = feature_infos_[i];
    if (fi == "none" || fi.empty()) continue;
    if (!first_fi) ss << ",";
    first_fi = false;
    ss << "\"" << feature_names_[i] << "\":{";
    if (fi.front() == '[') {
      auto colon = fi.find(':');
      ss << "\"min_value\":" << fi.substr(1, colon - 1) << ",\"max_value\":"
         << fi.substr(colon + 1, fi.size() - colon - 2) << ",\"values\":[]";
    } else {
      ss << "\"min_value\":0,\"max_value\":0,\"values\":[";
      auto cats = Common::SplitAny(fi.c_str(), ":");
      for (size_t c = 0; c < cats.size(); ++c) {
        if (c) ss << ",";
        ss << cats[c];
      }
      ss << "]";
    }
    ss << "}";
  }
  ss << "},";
  ss << "\"tree_info\":[";
  for (int t = start_iter * num_tree_per_iteration_;
       t < end_iter * num_tree_per_iteration_ && t < static_cast<int>(models_.size()); ++t) {
    if (t > start_iter * num_tree_per_iteration_) ss << ",";
    ss << "{\"tree_index\":" << t << ","
       << models_[t]->ToJSON().substr(1);  // merge tree fields
  }
  ss << "],";
  auto imp = FeatureImportance(num_iter, feature_importance_type);
  ss << "\"feature_importances\":{";
  bool first = true;
  for (int i = 0; i <= max_feature_idx_; ++i) {
    if (imp[i] > 0) {
      if (!first) ss << ",";
      first = false;
      std::string name = i < static_cast<int>(feature_names_.size())
                             ? feature_names_[i] : "Column_" + std::to_string(i);
      ss << "\"" << name << "\":" << imp[i];
    }
  }
  ss << "}}";
  return ss.str();
}

GBDT* GBDT::CreateBoosting(const std::string& type, const char* model_filename) {
  GBDT* ret = nullptr;
  if (type == "gbdt" || type == "goss") ret = new GBDT();
  else if (type == "dart") ret = new DART();
  else if (type == "rf") ret = new RF();
  else Log::Fatal("Unknown boosting type %s", type.c_str());
  if (model_filename != nullptr && model_filename[0] != '\0') {
    FILE* fp = fopen(model_filename, "rb");
    if (!fp) Log::Fatal("Model file %s not found", model_filename);
    fseek(fp, 0, SEEK_END);
    long sz = ftell(fp);
    fseek(fp, 0, SEEK_SET);
    std::string buf(sz, '\0');
    MIGBM_CHECK_EQ(fread(&buf[0], 1, sz, fp), static_cast<size_t>(sz));
    fclose(fp);
    ret->LoadModelFromString(buf.data(), buf.size());
  }
  return ret;
}


std::string GBDT::ModelToIfElse(int num_iteration) const {
  // Self-contained C++ source for the whole model (reference parity:
  // convert_model task with convert_model_language=cpp). Compiles with any
  // C++11 compiler; only <cmath> is required.
  int total_iters = num_tree_per_iteration_ > 0
                        ? static_cast<int>(models_.size()) / num_tree_per_iteration_ : 0;
  int end_iter = num_iteration > 0 ? std::min(num_iteration, total_iters) : total_iters;
  const int n_models = end_iter * num_tree_per_iteration_;
  std::stringstream ss;
  ss << std::setprecision(17);
  ss << "// Generated by migbm convert_model (if-else codegen).\n"
     << "// objective: " << objective_tostring_ << "\n"
     << "#include <cmath>\n\n"
     << "#define MIGBM_NUM_FEATURES " << (max_feature_idx_ + 1) << "\n"
     << "#define MIGBM_NUM_CLASSES " << num_class_ << "\n"
     << "#define MIGBM_NUM_TREES " << n_models << "\n\n"
     << "static inline bool NumericalDecision(double v, int missing_type, bool default_left,\n"
     << "                                     double threshold) {\n"
     << "  if (std::isnan(v) && missing_type != 2) v = 0.0;\n"
     << "  if ((missing_type == 1 && v == 0.0) || (missing_type == 2 && std::isnan(v)))\n"
     << "    return default_left;\n"
     << "  return v <= threshold;\n"
     << "}\n\n"
     << "static inline bool CategoricalDecision(double v, const unsigned int* bits, int n_words) {\n"
     << "  if (std::isnan(v)) return false;\n"
     << "  int c = static_cast<int>(v);\n"
     << "  if (c < 0) return false;\n"
     << "  return (c >> 5) < n_words && ((bits[c >> 5] >> (c & 31)) & 1U);\n"
     << "}\n\n";
  for (int i = 0; i < n_models; ++i) ss << models_[i]->ToIfElse(i);
  // raw score per class
  ss << "void PredictRaw(const double* features, double* output) {\n";
  for (int k = 0; k < num_tree_per_iteration_; ++k) {
    ss << "  output[" << k << "] = 0.0";
    for (int i = k; i < n_models; i += num_tree_per_iteration_)
      ss << "\n      + PredictTree" << i << "(features)";
    ss << ";\n";
  }
  if (average_output_ && end_iter > 0) {
    for (int k = 0; k < num_tree_per_iteration_; ++k)
      ss << "  output[" << k << "] /= " << end_iter << ".0;\n";
  }
  ss << "}\n\n";
  // transformed prediction matching the training objective
  const std::string& obj = objective_name_;
  double sigmoid = 1.0;
  {
    size_t p = objective_tostring_.find("sigmoid:");
    if (p != std::string::npos) sigmoid = atof(objective_tostring_.c_str() + p + 8);
  }
  ss << "void Predict(const double* features, double* output) {\n"
     << "  PredictRaw(features, output);\n";
  if (obj == "binary" || obj == "cross_entropy" || obj == "xentropy" ||
      obj == "multiclassova" || obj == "ova") {
    ss << "  for (int k = 0; k < MIGBM_NUM_CLASSES; ++k)\n"
       << "    output[k] = 1.0 / (1.0 + std::exp(-" << sigmoid << " * output[k]));\n";
  } else if (obj == "multiclass" || obj == "softmax") {
    ss << "  double wmax = output[0];\n"
       << "  for (int k = 1; k < MIGBM_NUM_CLASSES; ++k) if (output[k] > wmax) wmax = output[k];\n"
       << "  double wsum = 0.0;\n"
       << "  for (int k = 0; k < MIGBM_NUM_CLASSES; ++k) { output[k] = std::exp(output[k] - wmax); wsum += output[k]; }\n"
       << "  for (int k = 0; k < MIGBM_NUM_CLASSES; ++k) output[k] /= wsum;\n";
  } else if (obj == "poisson" || obj == "gamma" || obj == "tweedie") {
    ss << "  for (int k = 0; k < MIGBM_NUM_CLASSES; ++k) output[k] = std::exp(output[k]);\n";
  }
  ss << "}\n\n";
  // leaf indices (refit / feature engineering)
  ss << "void PredictLeafIndex(const double* features, int* output) {\n";
  for (int i = 0; i < n_models; ++i)
    ss << "  output[" << i << "] = PredictTree" << i << "LeafIndex(features);\n";
  ss << "}\n";
  return ss.str();
}

}  // namespace migbm
