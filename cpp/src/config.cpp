/*! migbm Config implementation. Alias table mirrors reference config_auto.cpp semantics
 *  (names only). */
#include "migbm/config.h"

namespace migbm {

const std::unordered_map<std::string, std::string>& Config::alias_table() {
  static const std::unordered_map<std::string, std::string> tbl = {
      {"config_file", "config"},
      {"task_type", "task"},
      {"objective_type", "objective"}, {"app", "objective"}, {"application", "objective"},
      {"loss", "objective"},
      {"boosting_type", "boosting"}, {"boost", "boosting"},
      {"train", "data"}, {"train_data", "data"}, {"train_data_file", "data"}, {"data_filename", "data"},
      {"test", "valid"}, {"valid_data", "valid"}, {"valid_data_file", "valid"}, {"test_data", "valid"},
      {"test_data_file", "valid"}, {"valid_filenames", "valid"},
      {"num_iteration", "num_iterations"}, {"n_iter", "num_iterations"}, {"num_tree", "num_iterations"},
      {"num_trees", "num_iterations"}, {"num_round", "num_iterations"}, {"num_rounds", "num_iterations"},
      {"nrounds", "num_iterations"}, {"num_boost_round", "num_iterations"}, {"n_estimators", "num_iterations"},
      {"max_iter", "num_iterations"},
      {"shrinkage_rate", "learning_rate"}, {"eta", "learning_rate"},
      {"num_leaf", "num_leaves"}, {"max_leaves", "num_leaves"}, {"max_leaf", "num_leaves"},
      {"max_leaf_nodes", "num_leaves"},
      {"tree", "tree_learner"}, {"tree_type", "tree_learner"}, {"tree_learner_type", "tree_learner"},
      {"num_thread", "num_threads"}, {"nthread", "num_threads"}, {"nthreads", "num_threads"},
      {"n_jobs", "num_threads"},
      {"device", "device_type"},
      {"random_seed", "seed"}, {"random_state", "seed"},
      {"hist_pool_size", "histogram_pool_size"},
      {"min_data_per_leaf", "min_data_in_leaf"}, {"min_data", "min_data_in_leaf"},
      {"min_child_samples", "min_data_in_leaf"}, {"min_samples_leaf", "min_data_in_leaf"},
      {"min_sum_hessian_per_leaf", "min_sum_hessian_in_leaf"}, {"min_sum_hessian", "min_sum_hessian_in_leaf"},
      {"min_hessian", "min_sum_hessian_in_leaf"}, {"min_child_weight", "min_sum_hessian_in_leaf"},
      {"sub_row", "bagging_fraction"}, {"subsample", "bagging_fraction"}, {"bagging", "bagging_fraction"},
      {"pos_sub_row", "pos_bagging_fraction"}, {"pos_subsample", "pos_bagging_fraction"},
      {"pos_bagging", "pos_bagging_fraction"},
      {"neg_sub_row", "neg_bagging_fraction"}, {"neg_subsample", "neg_bagging_fraction"},
      {"neg_bagging", "neg_bagging_fraction"},
      {"subsample_freq", "bagging_freq"},
      {"bagging_fraction_seed", "bagging_seed"},
      {"sub_feature", "feature_fraction"}, {"colsample_bytree", "feature_fraction"},
      {"sub_feature_bynode", "feature_fraction_bynode"}, {"colsample_bynode", "feature_fraction_bynode"},
      {"extra_tree", "extra_trees"},
      {"early_stopping_rounds", "early_stopping_round"}, {"early_stopping", "early_stopping_round"},
      {"n_iter_no_change", "early_stopping_round"},
      {"max_tree_output", "max_delta_step"}, {"max_leaf_output", "max_delta_step"},
      {"reg_alpha", "lambda_l1"}, {"l1_regularization", "lambda_l1"},
      {"reg_lambda", "lambda_l2"}, {"lambda", "lambda_l2"}, {"l2_regularization", "lambda_l2"},
      {"min_split_gain", "min_gain_to_split"},
      {"rate_drop", "drop_rate"},
      {"topk", "top_k"},
      {"mc", "monotone_constraints"}, {"monotone_constraint", "monotone_constraints"},
      {"monotone_constraining_method", "monotone_constraints_method"}, {"mc_method", "monotone_constraints_method"},
      {"monotone_splits_penalty", "monotone_penalty"}, {"ms_penalty", "monotone_penalty"}, {"mc_penalty", "monotone_penalty"},
      {"feature_contrib", "feature_contri"}, {"fc", "feature_contri"}, {"fp", "feature_contri"},
      {"feature_penalty", "feature_contri"},
      {"fs", "forcedsplits_filename"}, {"forced_splits_filename", "forcedsplits_filename"},
      {"forced_splits_file", "forcedsplits_filename"}, {"forced_splits", "forcedsplits_filename"},
      {"verbose", "verbosity"},
      {"model_input", "input_model"}, {"model_in", "input_model"},
      {"model_output", "output_model"}, {"model_out", "output_model"},
      {"save_period", "snapshot_freq"},
      {"linear_trees", "linear_tree"},
      {"max_bins", "max_bin"},
      {"subsample_for_bin", "bin_construct_sample_cnt"},
      {"data_seed", "data_random_seed"},
      {"is_sparse", "is_enable_sparse"}, {"enable_sparse", "is_enable_sparse"}, {"sparse", "is_enable_sparse"},
      {"is_enable_bundle", "enable_bundle"}, {"bundle", "enable_bundle"},
      {"is_pre_partition", "pre_partition"},
      {"two_round_loading", "two_round"}, {"use_two_round_loading", "two_round"},
      {"has_header", "header"},
      {"label", "label_column"},
      {"weight", "weight_column"},
      {"group", "group_column"}, {"group_id", "group_column"}, {"query_column", "group_column"},
      {"query", "group_column"}, {"query_id", "group_column"},
      {"ignore_feature", "ignore_column"}, {"blacklist", "ignore_column"},
      {"cat_feature", "categorical_feature"}, {"categorical_column", "categorical_feature"},
      {"cat_column", "categorical_feature"}, {"categorical_features", "categorical_feature"},
      {"is_save_binary", "save_binary"}, {"is_save_binary_file", "save_binary"},
      {"is_predict_raw_score", "predict_raw_score"}, {"predict_rawscore", "predict_raw_score"},
      {"raw_score", "predict_raw_score"},
      {"is_predict_leaf_index", "predict_leaf_index"}, {"leaf_index", "predict_leaf_index"},
      {"contrib", "predict_contrib"},
      {"predict_result", "output_result"}, {"prediction_result", "output_result"},
      {"unbalance", "is_unbalance"}, {"unbalanced_sets", "is_unbalance"},
      {"num_classes", "num_class"},
      {"metrics", "metric"}, {"metric_types", "metric"},
      {"output_freq", "metric_freq"},
      {"training_metric", "is_provide_training_metric"}, {"is_training_metric", "is_provide_training_metric"},
      {"train_metric", "is_provide_training_metric"},
      {"ndcg_eval_at", "eval_at"}, {"ndcg_at", "eval_at"}, {"map_eval_at", "eval_at"}, {"map_at", "eval_at"},
      {"num_machine", "num_machines"},
      {"local_port", "local_listen_port"}, {"port", "local_listen_port"},
      {"machine_list_file", "machine_list_filename"}, {"machine_list", "machine_list_filename"},
      {"mlist", "machine_list_filename"},
      {"workers", "machines"}, {"nodes", "machines"},
      {"gpu_platform", "gpu_platform_id"},
      {"gpu_device", "gpu_device_id"},
      {"gpu_dp", "gpu_use_dp"},
      {"num_gpus", "num_gpu"},
  };
  return tbl;
}

std::string Config::ResolveAlias(const std::string& key) {
  auto& tbl = alias_table();
  auto it = tbl.find(key);
  return it == tbl.end() ? key : it->second;
}

std::unordered_map<std::string, std::string> Config::Str2Map(const char* params) {
  std::unordered_map<std::string, std::string> out;
  for (auto& tok : Common::SplitAny(params, " \t\n\r")) {
    auto pos = tok.find('=');
    if (pos == std::string::npos) continue;
    std::string k = Common::Trim(tok.substr(0, pos));
    std::string v = Common::Trim(tok.substr(pos + 1));
    if (!k.empty()) out[Common::ToLower(k)] = v;
  }
  return out;
}

namespace {
bool ParseBool(const std::string& v) {
  std::string s = Common::ToLower(v);
  return !(s == "false" || s == "0" || s == "-" || s.empty());
}
int ParseInt(const std::string& v) { return static_cast<int>(strtoll(v.c_str(), nullptr, 10)); }
double ParseDouble(const std::string& v) { return strtod(v.c_str(), nullptr); }
std::vector<int> ParseIntList(const std::string& v) {
  std::vector<int> out;
  Common::StringToArray<int>(v, ',', &out);
  return out;
}
std::vector<double> ParseDoubleList(const std::string& v) {
  std::vector<double> out;
  Common::StringToArray<double>(v, ',', &out);
  return out;
}
}  // namespace

void Config::Set(const std::unordered_map<std::string, std::string>& params_in) {
  // resolve aliases first (last writer wins deterministically by canonical key sort)
  std::map<std::string, std::string> params;
  for (auto& kv : params_in) params[ResolveAlias(kv.first)] = kv.second;

  for (auto& kv : params) {
    const std::string& k = kv.first;
    const std::string& v = kv.second;
    raw[k] = v;
    if (k == "task") task = v;
    else if (k == "objective") objective = Common::ToLower(v);
    else if (k == "boosting") boosting = Common::ToLower(v);
    else if (k == "data_sample_strategy") data_sample_strategy = Common::ToLower(v);
    else if (k == "data") data = v;
    else if (k == "valid") valid = Common::Split(v.c_str(), ',');
    else if (k == "num_iterations") num_iterations = ParseInt(v);
    else if (k == "learning_rate") learning_rate = ParseDouble(v);
    else if (k == "num_leaves") num_leaves = ParseInt(v);
    else if (k == "tree_learner") tree_learner = Common::ToLower(v);
    else if (k == "num_threads") num_threads = ParseInt(v);
    else if (k == "device_type") device_type = Common::ToLower(v);
    else if (k == "seed") seed = ParseInt(v);
    else if (k == "deterministic") deterministic = ParseBool(v);
    else if (k == "force_col_wise") force_col_wise = ParseBool(v);
    else if (k == "force_row_wise") force_row_wise = ParseBool(v);
    else if (k == "histogram_pool_size") histogram_pool_size = ParseDouble(v);
    else if (k == "max_depth") max_depth = ParseInt(v);
    else if (k == "min_data_in_leaf") min_data_in_leaf = ParseInt(v);
    else if (k == "min_sum_hessian_in_leaf") min_sum_hessian_in_leaf = ParseDouble(v);
    else if (k == "bagging_fraction") bagging_fraction = ParseDouble(v);
    else if (k == "pos_bagging_fraction") pos_bagging_fraction = ParseDouble(v);
    else if (k == "neg_bagging_fraction") neg_bagging_fraction = ParseDouble(v);
    else if (k == "bagging_freq") bagging_freq = ParseInt(v);
    else if (k == "bagging_seed") bagging_seed = ParseInt(v);
    else if (k == "bagging_by_query") bagging_by_query = ParseBool(v);
    else if (k == "feature_fraction") feature_fraction = ParseDouble(v);
    else if (k == "feature_fraction_bynode") feature_fraction_bynode = ParseDouble(v);
    else if (k == "feature_fraction_seed") feature_fraction_seed = ParseInt(v);
    else if (k == "extra_trees") extra_trees = ParseBool(v);
    else if (k == "extra_seed") extra_seed = ParseInt(v);
    else if (k == "early_stopping_round") early_stopping_round = ParseInt(v);
    else if (k == "early_stopping_min_delta") early_stopping_min_delta = ParseDouble(v);
    else if (k == "first_metric_only") first_metric_only = ParseBool(v);
    else if (k == "max_delta_step") max_delta_step = ParseDouble(v);
    else if (k == "lambda_l1") lambda_l1 = ParseDouble(v);
    else if (k == "lambda_l2") lambda_l2 = ParseDouble(v);
    else if (k == "linear_lambda") linear_lambda = ParseDouble(v);
    else if (k == "min_gain_to_split") min_gain_to_split = ParseDouble(v);
    else if (k == "drop_rate") drop_rate = ParseDouble(v);
    else if (k == "max_drop") max_drop = ParseInt(v);
    else if (k == "skip_drop") skip_drop = ParseDouble(v);
    else if (k == "xgboost_dart_mode") xgboost_dart_mode = ParseBool(v);
    else if (k == "uniform_drop") uniform_drop = ParseBool(v);
    else if (k == "drop_seed") drop_seed = ParseInt(v);
    else if (k == "top_rate") top_rate = ParseDouble(v);
    else if (k == "other_rate") other_rate = ParseDouble(v);
    else if (k == "min_data_per_group") min_data_per_group = ParseInt(v);
    else if (k == "max_cat_threshold") max_cat_threshold = ParseInt(v);
    else if (k == "cat_l2") cat_l2 = ParseDouble(v);
    else if (k == "cat_smooth") cat_smooth = ParseDouble(v);
    else if (k == "max_cat_to_onehot") max_cat_to_onehot = ParseInt(v);
    else if (k == "top_k") top_k = ParseInt(v);
    else if (k == "monotone_constraints") monotone_constraints = ParseIntList(v);
    else if (k == "monotone_constraints_method") monotone_constraints_method = v;
    else if (k == "monotone_penalty") monotone_penalty = ParseDouble(v);
    else if (k == "feature_contri") feature_contri = ParseDoubleList(v);
    else if (k == "forcedsplits_filename") forcedsplits_filename = v;
    else if (k == "refit_decay_rate") refit_decay_rate = ParseDouble(v);
    else if (k == "cegb_tradeoff") cegb_tradeoff = ParseDouble(v);
    else if (k == "cegb_penalty_split") cegb_penalty_split = ParseDouble(v);
    else if (k == "cegb_penalty_feature_lazy") cegb_penalty_feature_lazy = ParseDoubleList(v);
    else if (k == "cegb_penalty_feature_coupled") cegb_penalty_feature_coupled = ParseDoubleList(v);
    else if (k == "path_smooth") path_smooth = ParseDouble(v);
    else if (k == "interaction_constraints") interaction_constraints = v;
    else if (k == "verbosity") verbosity = ParseInt(v);
    else if (k == "input_model") input_model = v;
    else if (k == "convert_model_language") convert_model_language = v;
    else if (k == "convert_model") convert_model = v;
    else if (k == "output_model") output_model = v;
    else if (k == "snapshot_freq") snapshot_freq = ParseInt(v);
    else if (k == "use_quantized_grad") use_quantized_grad = ParseBool(v);
    else if (k == "num_grad_quant_bins") num_grad_quant_bins = ParseInt(v);
    else if (k == "quant_train_renew_leaf") quant_train_renew_leaf = ParseBool(v);
    else if (k == "stochastic_rounding") stochastic_rounding = ParseBool(v);
    else if (k == "linear_tree") linear_tree = ParseBool(v);
    else if (k == "max_bin") max_bin = ParseInt(v);
    else if (k == "max_bin_by_feature") max_bin_by_feature = ParseIntList(v);
    else if (k == "min_data_in_bin") min_data_in_bin = ParseInt(v);
    else if (k == "bin_construct_sample_cnt") bin_construct_sample_cnt = ParseInt(v);
    else if (k == "data_random_seed") data_random_seed = ParseInt(v);
    else if (k == "is_enable_sparse") is_enable_sparse = ParseBool(v);
    else if (k == "enable_bundle") enable_bundle = ParseBool(v);
    else if (k == "use_missing") use_missing = ParseBool(v);
    else if (k == "zero_as_missing") zero_as_missing = ParseBool(v);
    else if (k == "feature_pre_filter") feature_pre_filter = ParseBool(v);
    else if (k == "pre_partition") pre_partition = ParseBool(v);
    else if (k == "two_round") two_round = ParseBool(v);
    else if (k == "header") header = ParseBool(v);
    else if (k == "label_column") label_column = v;
    else if (k == "weight_column") weight_column = v;
    else if (k == "group_column") group_column = v;
    else if (k == "ignore_column") ignore_column = v;
    else if (k == "categorical_feature") categorical_feature = v;
    else if (k == "forcedbins_filename") { forcedbins_filename = v; forcedbins_filename_set = true; }
    else if (k == "save_binary") save_binary = ParseBool(v);
    else if (k == "max_conflict_rate") max_conflict_rate = ParseDouble(v);
    else if (k == "start_iteration_predict") start_iteration_predict = ParseInt(v);
    else if (k == "num_iteration_predict") num_iteration_predict = ParseInt(v);
    else if (k == "predict_raw_score") predict_raw_score = ParseBool(v);
    else if (k == "predict_leaf_index") predict_leaf_index = ParseBool(v);
    else if (k == "predict_contrib") predict_contrib = ParseBool(v);
    else if (k == "predict_disable_shape_check") predict_disable_shape_check = ParseBool(v);
    else if (k == "output_result") output_result = v;
    else if (k == "num_class") num_class = ParseInt(v);
    else if (k == "is_unbalance") is_unbalance = ParseBool(v);
    else if (k == "scale_pos_weight") scale_pos_weight = ParseDouble(v);
    else if (k == "sigmoid") sigmoid = ParseDouble(v);
    else if (k == "boost_from_average") boost_from_average = ParseBool(v);
    else if (k == "reg_sqrt") reg_sqrt = ParseBool(v);
    else if (k == "alpha") alpha = ParseDouble(v);
    else if (k == "fair_c") fair_c = ParseDouble(v);
    else if (k == "poisson_max_delta_step") poisson_max_delta_step = ParseDouble(v);
    else if (k == "tweedie_variance_power") tweedie_variance_power = ParseDouble(v);
    else if (k == "lambdarank_truncation_level") lambdarank_truncation_level = ParseInt(v);
    else if (k == "lambdarank_norm") lambdarank_norm = ParseBool(v);
    else if (k == "label_gain") label_gain = ParseDoubleList(v);
    else if (k == "objective_seed") objective_seed = ParseInt(v);
    else if (k == "metric") {
      metric.clear();
      for (auto& m : Common::Split(v.c_str(), ',')) {
        auto mm = Common::Trim(Common::ToLower(m));
        if (!mm.empty()) metric.push_back(mm);
      }
    }
    else if (k == "metric_freq") metric_freq = ParseInt(v);
    else if (k == "is_provide_training_metric") is_provide_training_metric = ParseBool(v);
    else if (k == "eval_at") eval_at = ParseIntList(v);
    else if (k == "multi_error_top_k") multi_error_top_k = ParseInt(v);
    else if (k == "auc_mu_weights") auc_mu_weights = ParseDoubleList(v);
    else if (k == "num_machines") num_machines = ParseInt(v);
    else if (k == "local_listen_port") local_listen_port = ParseInt(v);
    else if (k == "time_out") time_out = ParseInt(v);
    else if (k == "machine_list_filename") machine_list_filename = v;
    else if (k == "machines") machines = v;
    else if (k == "gpu_platform_id") gpu_platform_id = ParseInt(v);
    else if (k == "gpu_device_id") gpu_device_id = ParseInt(v);
    else if (k == "gpu_use_dp") gpu_use_dp = ParseBool(v);
    else if (k == "num_gpu") num_gpu = ParseInt(v);
    else if (k == "gpu_device_id_list") gpu_device_id_list = v;
    else if (k == "saved_feature_importance_type") saved_feature_importance_type = ParseInt(v);
    else if (k == "precise_float_parser") precise_float_parser = ParseBool(v);
    else if (k == "parser_config_file") parser_config_file = v;
    else if (k == "pred_early_stop") pred_early_stop = ParseBool(v);
    else if (k == "pred_early_stop_freq") pred_early_stop_freq = ParseInt(v);
    else if (k == "pred_early_stop_margin") pred_early_stop_margin = ParseDouble(v);
    else if (k == "lambdarank_position_bias_regularization")
      lambdarank_position_bias_regularization = ParseDouble(v);
    // unknown keys are kept in raw only (tolerated, like the reference's pass-through)
  }

  // master `seed` cascades into every sub-seed not explicitly given
  // (parity: reference src/io/config.cpp Config::Set seed handling)
  if (params.count("seed")) {
    Random rand(seed);
    const int int_max = 32767;
    if (!params.count("data_random_seed")) data_random_seed = rand.NextInt(0, int_max);
    if (!params.count("bagging_seed")) bagging_seed = rand.NextInt(0, int_max);
    if (!params.count("drop_seed")) drop_seed = rand.NextInt(0, int_max);
    if (!params.count("feature_fraction_seed")) feature_fraction_seed = rand.NextInt(0, int_max);
    if (!params.count("objective_seed")) objective_seed = rand.NextInt(0, int_max);
    if (!params.count("extra_seed")) extra_seed = rand.NextInt(0, int_max);
  }

  // objective aliases
  if (objective == "regression_l2" || objective == "mean_squared_error" || objective == "mse" ||
      objective == "l2_root" || objective == "root_mean_squared_error" || objective == "rmse")
    objective = "regression";
  if (objective == "l2") objective = "regression";
  if (objective == "mean_absolute_error" || objective == "mae" || objective == "l1") objective = "regression_l1";
  if (objective == "mean_absolute_percentage_error") objective = "mape";
  if (objective == "lambdarank") objective = "lambdarank";
  if (objective == "rank_xendcg" || objective == "xendcg" || objective == "xe_ndcg" ||
      objective == "xe_ndcg_mart" || objective == "xendcg_mart")
    objective = "rank_xendcg";
  if (objective == "multiclassova" || objective == "multiclass_ova" || objective == "ova" || objective == "ovr")
    objective = "multiclassova";
  if (objective == "softmax") objective = "multiclass";
  if (objective == "cross_entropy" || objective == "xentropy") objective = "cross_entropy";
  if (objective == "cross_entropy_lambda" || objective == "xentlambda") objective = "cross_entropy_lambda";

  // boosting=goss back-compat -> gbdt + goss sampling
  if (boosting == "goss") {
    boosting = "gbdt";
    data_sample_strategy = "goss";
  }
  if (boosting == "gbrt") boosting = "gbdt";
  if (boosting == "random_forest") boosting = "rf";

  if (deterministic) {
    // thread-count-invariant training: col-wise histograms accumulate each
    // feature sequentially, and the root reduction uses fixed-block summation
    force_col_wise = true;
    force_row_wise = false;
  }
  if (linear_tree) {
    // reference restrictions: per-leaf linear fits need the serial learner's raw
    // values and exact partitions
    if (tree_learner != "serial") {
      tree_learner = "serial";
      Log::Warning("Linear tree learner must be serial; forcing tree_learner=serial");
    }
    if (zero_as_missing)
      Log::Fatal("zero_as_missing must be false when fitting linear trees");
  }
  // intermediate/advanced monotone constraints need full local histograms and
  // stable per-node feature sets (reference CheckParamConflict, config.cpp:449-458)
  if (monotone_constraints_method == "intermediate" ||
      monotone_constraints_method == "advanced") {
    if (tree_learner != "serial" && num_machines > 1) {
      Log::Warning("Cannot use intermediate/advanced monotone constraints in "
                   "distributed learning; falling back to method=basic");
      monotone_constraints_method = "basic";
    } else if (feature_fraction_bynode != 1.0) {
      Log::Warning("feature_fraction_bynode is incompatible with intermediate/"
                   "advanced monotone constraints; falling back to method=basic");
      monotone_constraints_method = "basic";
    }
  }
  if (num_threads > 0) omp_set_num_threads(num_threads);
  // only an EXPLICIT verbosity touches the global log level: lazily-constructed
  // datasets with default params must not undo a booster's verbosity=-1
  if (params.count("verbosity")) {
    if (verbosity <= -1) Log::Level() = LogLevel::Fatal;
    else if (verbosity == 0) Log::Level() = LogLevel::Warning;
    else if (verbosity == 1) Log::Level() = LogLevel::Info;
    else Log::Level() = LogLevel::Debug;
  }
}

std::string Config::SaveHyperParameters() const {
  std::stringstream ss;
  ss << "parameters:" << '\n';
  // echo canonical values for key params (model-file "parameters:" block)
  ss << "[boosting: " << boosting << "]\n";
  ss << "[objective: " << objective << "]\n";
  ss << "[metric: " << Common::Join(metric, ",") << "]\n";
  ss << "[tree_learner: " << tree_learner << "]\n";
  ss << "[device_type: " << device_type << "]\n";
  ss << "[data_sample_strategy: " << data_sample_strategy << "]\n";
  ss << "[data: " << data << "]\n";
  ss << "[valid: " << Common::Join(valid, ",") << "]\n";
  ss << "[num_iterations: " << num_iterations << "]\n";
  ss << "[learning_rate: " << learning_rate << "]\n";
  ss << "[num_leaves: " << num_leaves << "]\n";
  ss << "[num_threads: " << num_threads << "]\n";
  ss << "[seed: " << seed << "]\n";
  ss << "[deterministic: " << (deterministic ? 1 : 0) << "]\n";
  ss << "[max_depth: " << max_depth << "]\n";
  ss << "[min_data_in_leaf: " << min_data_in_leaf << "]\n";
  ss << "[min_sum_hessian_in_leaf: " << min_sum_hessian_in_leaf << "]\n";
  ss << "[bagging_fraction: " << bagging_fraction << "]\n";
  ss << "[bagging_freq: " << bagging_freq << "]\n";
  ss << "[bagging_seed: " << bagging_seed << "]\n";
  ss << "[feature_fraction: " << feature_fraction << "]\n";
  ss << "[feature_fraction_bynode: " << feature_fraction_bynode << "]\n";
  ss << "[feature_fraction_seed: " << feature_fraction_seed << "]\n";
  ss << "[extra_trees: " << (extra_trees ? 1 : 0) << "]\n";
  ss << "[early_stopping_round: " << early_stopping_round << "]\n";
  ss << "[max_delta_step: " << max_delta_step << "]\n";
  ss << "[lambda_l1: " << lambda_l1 << "]\n";
  ss << "[lambda_l2: " << lambda_l2 << "]\n";
  ss << "[min_gain_to_split: " << min_gain_to_split << "]\n";
  ss << "[max_bin: " << max_bin << "]\n";
  ss << "[min_data_in_bin: " << min_data_in_bin << "]\n";
  ss << "[bin_construct_sample_cnt: " << bin_construct_sample_cnt << "]\n";
  ss << "[data_random_seed: " << data_random_seed << "]\n";
  ss << "[num_class: " << num_class << "]\n";
  ss << "[sigmoid: " << sigmoid << "]\n";
  ss << "[boost_from_average: " << (boost_from_average ? 1 : 0) << "]\n";
  ss << "[num_machines: " << num_machines << "]\n";
  ss << "[num_gpu: " << num_gpu << "]\n";
  // any extra raw params the user set that aren't canonicalized above
  ss << "end of parameters" << '\n';
  return ss.str();
}

}  // namespace migbm
