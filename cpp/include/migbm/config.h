/*!
 * migbm Config — typed hyper-parameters with LightGBM-compatible names and aliases.
 * Capability parity target: reference include/LightGBM/config.h (+config_auto.cpp alias
 * tables). Fresh implementation; only the externally visible names/aliases/defaults match.
 */
#ifndef MIGBM_CONFIG_H_
#define MIGBM_CONFIG_H_

#include "common.h"

#include <map>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

namespace migbm {

enum class TaskType { kTrain, kPredict, kConvertModel, kRefitTree };

struct Config {
  // ---- core
  std::string task = "train";
  std::string objective = "regression";
  std::string boosting = "gbdt";            // gbdt, dart, rf, goss(alias path)
  std::string data_sample_strategy = "bagging";  // bagging, goss
  std::string data = "";
  std::vector<std::string> valid;
  int num_iterations = 100;                  // aliases: n_estimators, num_trees, ...
  double learning_rate = 0.1;
  int num_leaves = 31;
  std::string tree_learner = "serial";       // serial, feature, data, voting
  int num_threads = 0;
  std::string device_type = "cpu";           // cpu, gpu, cuda  (gpu/cuda -> HIP learner)
  int seed = 0;                              // master seed
  bool deterministic = false;

  // ---- learning control
  bool force_col_wise = false;
  bool force_row_wise = false;
  double histogram_pool_size = -1.0;
  int max_depth = -1;
  int min_data_in_leaf = 20;
  double min_sum_hessian_in_leaf = 1e-3;
  double bagging_fraction = 1.0;
  double pos_bagging_fraction = 1.0;
  double neg_bagging_fraction = 1.0;
  int bagging_freq = 0;
  int bagging_seed = 3;
  bool bagging_by_query = false;
  double feature_fraction = 1.0;
  double feature_fraction_bynode = 1.0;
  int feature_fraction_seed = 2;
  bool extra_trees = false;
  int extra_seed = 6;
  int early_stopping_round = 0;
  double early_stopping_min_delta = 0.0;
  bool first_metric_only = false;
  double max_delta_step = 0.0;
  double lambda_l1 = 0.0;
  double lambda_l2 = 0.0;
  double linear_lambda = 0.0;
  double min_gain_to_split = 0.0;
  double drop_rate = 0.1;                    // dart
  int max_drop = 50;
  double skip_drop = 0.5;
  bool xgboost_dart_mode = false;
  bool uniform_drop = false;
  int drop_seed = 4;
  double top_rate = 0.2;                     // goss
  double other_rate = 0.1;
  int min_data_per_group = 100;
  int max_cat_threshold = 32;
  double cat_l2 = 10.0;
  double cat_smooth = 10.0;
  int max_cat_to_onehot = 4;
  int top_k = 20;                            // voting parallel
  std::vector<int> monotone_constraints;
  std::string monotone_constraints_method = "basic";
  double monotone_penalty = 0.0;
  std::vector<double> feature_contri;
  std::string forcedsplits_filename = "";
  double refit_decay_rate = 0.9;
  double cegb_tradeoff = 1.0;
  double cegb_penalty_split = 0.0;
  std::vector<double> cegb_penalty_feature_lazy;
  std::vector<double> cegb_penalty_feature_coupled;
  double path_smooth = 0.0;
  std::string interaction_constraints = "";
  int verbosity = 1;
  std::string input_model = "";
  std::string convert_model_language = "";        // "" (json) or "cpp"
  std::string convert_model = "gbdt_prediction.cpp";
  std::string output_model = "LightGBM_model.txt";
  int snapshot_freq = -1;
  bool use_quantized_grad = false;
  int num_grad_quant_bins = 4;
  bool quant_train_renew_leaf = false;
  bool stochastic_rounding = true;

  // ---- dataset
  bool linear_tree = false;
  int max_bin = 255;
  std::vector<int> max_bin_by_feature;
  int min_data_in_bin = 3;
  int bin_construct_sample_cnt = 200000;
  int data_random_seed = 1;
  bool is_enable_sparse = true;
  bool enable_bundle = true;                 // EFB
  bool use_missing = true;
  bool zero_as_missing = false;
  bool feature_pre_filter = true;
  bool pre_partition = false;
  bool two_round = false;
  bool header = false;
  std::string label_column = "";
  std::string weight_column = "";
  std::string group_column = "";
  std::string ignore_column = "";
  std::string categorical_feature = "";
  bool forcedbins_filename_set = false;
  std::string forcedbins_filename = "";
  bool save_binary = false;
  double max_conflict_rate = 0.0;            // EFB conflict tolerance (0 = none)

  // ---- predict
  int start_iteration_predict = 0;
  int num_iteration_predict = -1;
  bool predict_raw_score = false;
  bool predict_leaf_index = false;
  bool predict_contrib = false;
  bool predict_disable_shape_check = false;
  std::string output_result = "LightGBM_predict_result.txt";

  // ---- objective
  int num_class = 1;
  bool is_unbalance = false;
  double scale_pos_weight = 1.0;
  double sigmoid = 1.0;
  bool boost_from_average = true;
  bool reg_sqrt = false;
  double alpha = 0.9;                        // huber/quantile
  double fair_c = 1.0;
  double poisson_max_delta_step = 0.7;
  double tweedie_variance_power = 1.5;
  int lambdarank_truncation_level = 30;
  bool lambdarank_norm = true;
  double lambdarank_position_bias_regularization = 0.0;
  std::vector<double> label_gain;
  int objective_seed = 5;

  // ---- metric
  std::vector<std::string> metric;
  int metric_freq = 1;
  bool is_provide_training_metric = false;
  std::vector<int> eval_at;                  // ndcg@/map@ positions
  int multi_error_top_k = 1;
  std::vector<double> auc_mu_weights;

  // ---- network
  int num_machines = 1;
  int local_listen_port = 12400;
  int time_out = 120;
  std::string machine_list_filename = "";
  std::string machines = "";

  // ---- device
  int gpu_platform_id = -1;
  int gpu_device_id = -1;
  bool gpu_use_dp = false;
  int num_gpu = 1;
  std::string gpu_device_id_list;  // comma-separated device ids for num_gpu>1
  // model io / loader / prediction knobs (reference parameter_set parity)
  int saved_feature_importance_type = 0;  // 0 split, 1 gain (CLI model save)
  bool precise_float_parser = false;      // our parser is always strtod-precise
  std::string parser_config_file;
  bool pred_early_stop = false;
  int pred_early_stop_freq = 10;
  double pred_early_stop_margin = 10.0;

  // raw key->value as given by user (post-alias-resolution), echoed into model file
  std::map<std::string, std::string> raw;

  // ------------------------------------------------------------------
  void Set(const std::unordered_map<std::string, std::string>& params);

  static const std::unordered_map<std::string, std::string>& alias_table();
  static const std::set<std::string>& parameter_set();

  /*! Parse "k1=v1 k2=v2" strings (also handles newline/tab separation). */
  static std::unordered_map<std::string, std::string> Str2Map(const char* params);

  std::string SaveHyperParameters() const;

  /*! Resolve an alias to canonical name (returns input if unknown). */
  static std::string ResolveAlias(const std::string& key);
};

}  // namespace migbm

#endif  // MIGBM_CONFIG_H_
