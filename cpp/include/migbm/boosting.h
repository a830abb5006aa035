/*!
 * migbm Boosting engine — GBDT / DART / RF over any TreeLearner.
 * Capability parity target: reference include/LightGBM/boosting.h, src/boosting/gbdt.{h,cpp},
 * gbdt_model_text.cpp, dart.hpp, rf.hpp, bagging.hpp, goss.hpp. Fresh implementation.
 */
#ifndef MIGBM_BOOSTING_H_
#define MIGBM_BOOSTING_H_

#include "config.h"
#include "dataset.h"
#include "metric.h"
#include "objective.h"
#include "tree.h"
#include "tree_learner.h"

#include <memory>
#include <string>
#include <vector>

namespace migbm {

/*! Bagging / GOSS row-sampling strategy (parity: sample_strategy.cpp / bagging.hpp / goss.hpp). */
class SampleStrategy {
 public:
  virtual ~SampleStrategy() = default;
  virtual void Bagging(int iter, TreeLearner* learner, score_t* gradients,
                       score_t* hessians) = 0;
  /*! true when Bagging() reads/modifies the gradient arrays (GOSS): device
   *  boosting must fall back to host gradients for such strategies. */
  virtual bool NeedsGradients() const { return false; }
  const std::vector<data_size_t>& bag_indices() const { return bag_indices_; }
  data_size_t bag_cnt() const { return bag_cnt_; }
  bool is_use_subset() const { return false; }
  static SampleStrategy* Create(const Config* cfg, const Dataset* data,
                                const ObjectiveFunction* obj, int num_tree_per_iter);

 protected:
  std::vector<data_size_t> bag_indices_;
  data_size_t bag_cnt_ = 0;
};

class GBDT {
 public:
  GBDT() = default;
  virtual ~GBDT() {
    if (Timer::Enabled()) Timer::Global().Print();
  }

  virtual void Init(const Config* config, const Dataset* train_data,
                    const ObjectiveFunction* objective,
                    const std::vector<const Metric*>& training_metrics);

  virtual void AddValidDataset(const Dataset* valid_data,
                               const std::vector<const Metric*>& valid_metrics);

  /*! one boosting iteration; custom grad/hess may be passed (nullptr = use objective). */
  virtual bool TrainOneIter(const score_t* gradients, const score_t* hessians);
  virtual void RollbackOneIter();
  virtual void Train(int snapshot_freq, const std::string& model_output_path);

  int GetCurrentIteration() const { return static_cast<int>(models_.size()) / num_tree_per_iteration_; }
  virtual void ResetTrainingData(const Dataset* train_data, const ObjectiveFunction* objective,
                                 const std::vector<const Metric*>& training_metrics);
  virtual void ResetConfig(const Config* config);

  /*! evaluation */
  std::vector<double> GetEvalAt(int data_idx) const;  // 0 = train, 1.. = valid
  std::vector<std::string> EvalNames() const;
  virtual void MergeFrom(const GBDT* other);

  /*! raw-feature single-row prediction */
  void PredictRaw(const double* features, double* output, int start_iter, int num_iter) const;
  /*! margin-based early stop across trees (parity: prediction_early_stop.cpp) */
  void PredictRawEarlyStop(const double* features, double* output, int start_iter,
                           int num_iter, int round_period, double margin_threshold,
                           bool multiclass) const;
  void ConvertRawToOutput(double* output) const {
    if (objective_ != nullptr) objective_->ConvertOutput(output, output);
  }
  void Predict(const double* features, double* output, int start_iter, int num_iter) const;
  void PredictLeafIndex(const double* features, double* output, int start_iter,
                        int num_iter) const;
  void PredictContrib(const double* features, double* output, int start_iter,
                      int num_iter) const;

  /*! training-data score access (for refit / custom logic) */
  const double* GetTrainingScore(int64_t* out_len) const;
  int64_t GetNumPredictAt(int data_idx) const;
  void GetPredictAt(int data_idx, double* result, int64_t* out_len) const;

  virtual bool EvalAndCheckEarlyStopping();

  std::string SaveModelToString(int start_iter, int num_iter, int feature_importance_type) const;
  bool SaveModelToFile(int start_iter, int num_iter, int feature_importance_type,
                       const char* filename) const;
  std::string DumpModel(int start_iter, int num_iter, int feature_importance_type) const;
  /*! whole model as a self-contained C++ source file (if-else codegen). */
  std::string ModelToIfElse(int num_iteration) const;
  bool LoadModelFromString(const char* str, size_t len);

  std::vector<double> FeatureImportance(int num_iter, int importance_type) const;

  int NumberOfTotalModel() const { return static_cast<int>(models_.size()); }
  int num_tree_per_iteration() const { return num_tree_per_iteration_; }
  int num_class() const { return num_class_; }
  int MaxFeatureIdx() const { return max_feature_idx_; }
  int LabelIdx() const { return label_idx_; }
  const std::vector<std::string>& FeatureNames() const { return feature_names_; }
  const std::string& ObjectiveName() const { return objective_name_; }
  const std::string& LoadedParameter() const { return loaded_parameter_; }
  const Tree* GetTree(int i) const { return models_[i].get(); }
  /*! shuffle tree order in [start, end) iterations (reference ShuffleModels). */
  void ShuffleModels(int start_iter, int end_iter) {
    const int total = GetCurrentIteration();
    start_iter = std::max(0, start_iter);
    end_iter = end_iter <= 0 ? total : std::min(end_iter, total);
    Random rng(17);
    for (int i = start_iter; i < end_iter - 1; ++i) {
      const int j = i + rng.NextInt(0, end_iter - i);
      for (int c = 0; c < num_tree_per_iteration_; ++c)
        std::swap(models_[static_cast<size_t>(i) * num_tree_per_iteration_ + c],
                  models_[static_cast<size_t>(j) * num_tree_per_iteration_ + c]);
    }
  }
  Tree* GetMutableTree(int i) { return models_[i].get(); }
  double GetLeafValue(int tree_idx, int leaf_idx) const {
    return models_[tree_idx]->LeafOutput(leaf_idx);
  }
  void SetLeafValue(int tree_idx, int leaf_idx, double v) {
    models_[tree_idx]->SetLeafOutput(leaf_idx, v);
  }
  double GetUpperBoundValue() const;
  double GetLowerBoundValue() const;
  virtual const char* SubModelName() const { return "tree"; }
  virtual bool IsLinear() const { return false; }
  bool average_output() const { return average_output_; }
  int NumPredictOneRow(int start_iter, int num_iter, bool predict_leaf, bool contrib) const;

  /*! refit leaf outputs from leaf predictions of existing structure */
  void RefitTree(const int32_t* leaf_preds, int nrow, int ncol);

  static GBDT* CreateBoosting(const std::string& type, const char* model_filename);

 protected:
  virtual void UpdateScore(const Tree* tree, int cur_tree_id);
  virtual double BoostFromAverage(int class_id, bool update_scores);
  virtual bool GetIsConstHessian() const {
    return objective_ != nullptr && objective_->IsConstantHessian();
  }
  std::string OutputMetric(int iter);

  const Config* config_ = nullptr;
  Config config_store_;
  std::unique_ptr<TreeLearner> tree_learner_;
  const Dataset* train_data_ = nullptr;
  const ObjectiveFunction* objective_ = nullptr;
  std::string objective_name_;
  std::string objective_tostring_;
  std::vector<const Metric*> training_metrics_;
  std::vector<const Dataset*> valid_data_;
  std::vector<std::vector<const Metric*>> valid_metrics_;

  int num_tree_per_iteration_ = 1;
  int num_class_ = 1;
  int label_idx_ = 0;
  int max_feature_idx_ = 0;
  int iter_ = 0;
  data_size_t num_data_ = 0;
  double shrinkage_rate_ = 0.1;
  bool average_output_ = false;
  std::vector<std::string> feature_names_;
  std::vector<std::string> feature_infos_;  // loaded-model bin info strings

  std::vector<std::unique_ptr<Tree>> models_;
  std::vector<double> train_score_;            // num_data * num_class, class-major
  std::vector<std::vector<double>> valid_score_;
  std::vector<score_t> gradients_, hessians_;
  std::unique_ptr<SampleStrategy> sample_strategy_;
  std::vector<double> init_scores_;            // per class (boost_from_average)
  // early stopping bookkeeping
  std::vector<double> best_metric_;
  int best_iter_ = 0;
  int es_rounds_since_best_ = 0;
  std::vector<int> es_counts_;
  // loaded-model state
  std::string loaded_parameter_;
  std::unique_ptr<const ObjectiveFunction> loaded_objective_;
};

class DART : public GBDT {
 public:
  bool TrainOneIter(const score_t* gradients, const score_t* hessians) override;
  const char* SubModelName() const override { return "tree"; }

 private:
  std::vector<int> DroppingTrees();
  void Normalize(const std::vector<int>& dropped, Tree* new_tree, int cur_tree_id);
  Random drop_rng_{4};
  bool drop_rng_init_ = false;
  std::vector<double> tree_weight_;  // per iteration; drives weighted (non-uniform) drop
  double sum_weight_ = 0.0;
};

class RF : public GBDT {
 public:
  void Init(const Config* config, const Dataset* train_data, const ObjectiveFunction* objective,
            const std::vector<const Metric*>& training_metrics) override;
  bool TrainOneIter(const score_t* gradients, const score_t* hessians) override;
};

}  // namespace migbm

#endif  // MIGBM_BOOSTING_H_
