/*! migbm GBDT boosting engine + bagging/GOSS sampling + DART + RF.
 *  Parity target: reference src/boosting/gbdt.cpp (TrainOneIter, UpdateScore,
 *  BoostFromAverage), bagging.hpp, goss.hpp, dart.hpp, rf.hpp. */
#include "migbm/boosting.h"
#include "migbm/network.h"

#include <algorithm>
#include <numeric>

namespace migbm {

// ------------------------------------------------------------------ sampling
namespace {

class BaggingStrategy : public SampleStrategy {
 public:
  BaggingStrategy(const Config* cfg, const Dataset* data)
      : cfg_(cfg), data_(data), rng_(cfg->bagging_seed) {}
  void Bagging(int iter, TreeLearner* learner, score_t*, score_t*) override {
    const bool need = cfg_->bagging_freq > 0 &&
                      (cfg_->bagging_fraction < 1.0 || cfg_->pos_bagging_fraction < 1.0 ||
                       cfg_->neg_bagging_fraction < 1.0);
    if (!need) { bag_cnt_ = 0; bag_indices_.clear(); learner->SetBaggingData(nullptr, nullptr, 0); return; }
    if (iter % cfg_->bagging_freq != 0 && !bag_indices_.empty()) {
      // keep previous bag
      learner->SetBaggingData(nullptr, bag_indices_.data(), bag_cnt_);
      return;
    }
    const data_size_t n = data_->num_data();
    bag_indices_.clear();
    bag_indices_.reserve(n);
    const bool posneg = cfg_->pos_bagging_fraction < 1.0 || cfg_->neg_bagging_fraction < 1.0;
    const label_t* label = data_->metadata().label();
    const data_size_t* qb = data_->metadata().query_boundaries();
    if (cfg_->bagging_by_query && qb != nullptr) {
      // sample whole queries: keeps ranking groups intact in the bag
      const data_size_t nq = data_->metadata().num_queries();
      for (data_size_t q = 0; q < nq; ++q) {
        if (rng_.NextFloat() < cfg_->bagging_fraction)
          for (data_size_t i = qb[q]; i < qb[q + 1]; ++i) bag_indices_.push_back(i);
      }
    } else {
      for (data_size_t i = 0; i < n; ++i) {
        double frac = cfg_->bagging_fraction;
        if (posneg) frac = label[i] > 0 ? cfg_->pos_bagging_fraction : cfg_->neg_bagging_fraction;
        if (rng_.NextFloat() < frac) bag_indices_.push_back(i);
      }
    }
    bag_cnt_ = static_cast<data_size_t>(bag_indices_.size());
    learner->SetBaggingData(nullptr, bag_indices_.data(), bag_cnt_);
  }

 private:
  const Config* cfg_;
  const Dataset* data_;
  Random rng_;
};

/*! Gradient one-side sampling (parity: goss.hpp). */
class GOSSStrategy : public SampleStrategy {
 public:
  GOSSStrategy(const Config* cfg, const Dataset* data, int num_tree_per_iter)
      : cfg_(cfg), data_(data), ntpi_(num_tree_per_iter), rng_(cfg->bagging_seed) {}
  bool NeedsGradients() const override { return true; }
  void Bagging(int iter, TreeLearner* learner, score_t* gradients,
               score_t* hessians) override {
    (void)iter;
    const data_size_t n = data_->num_data();
    // |g*h| magnitude
    std::vector<std::pair<float, data_size_t>> mag(n);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < n; ++i) {
      float m = 0;
      for (int c = 0; c < ntpi_; ++c) {
        size_t k = static_cast<size_t>(c) * n + i;
        m += std::fabs(gradients[k] * hessians[k]);
      }
      mag[i] = {m, i};
    }
    data_size_t top_k = static_cast<data_size_t>(n * cfg_->top_rate);
    data_size_t other_k = static_cast<data_size_t>(n * cfg_->other_rate);
    top_k = std::max<data_size_t>(1, top_k);
    std::partial_sort(mag.begin(), mag.begin() + top_k, mag.end(),
                      [](auto& a, auto& b) { return a.first > b.first; });
    const double multiply = static_cast<double>(n - top_k) / std::max<data_size_t>(1, other_k);
    bag_indices_.clear();
    for (data_size_t i = 0; i < top_k; ++i) bag_indices_.push_back(mag[i].second);
    // uniform sample of the rest
    for (data_size_t i = top_k; i < n; ++i) {
      if (rng_.NextFloat() < static_cast<double>(other_k) / (n - top_k)) {
        data_size_t r = mag[i].second;
        bag_indices_.push_back(r);
        for (int c = 0; c < ntpi_; ++c) {
          size_t k = static_cast<size_t>(c) * n + r;
          gradients[k] = static_cast<score_t>(gradients[k] * multiply);
          hessians[k] = static_cast<score_t>(hessians[k] * multiply);
        }
      }
    }
    std::sort(bag_indices_.begin(), bag_indices_.end());
    bag_cnt_ = static_cast<data_size_t>(bag_indices_.size());
    learner->SetBaggingData(nullptr, bag_indices_.data(), bag_cnt_);
  }

 private:
  const Config* cfg_;
  const Dataset* data_;
  int ntpi_;
  Random rng_;
};

}  // namespace

SampleStrategy* SampleStrategy::Create(const Config* cfg, const Dataset* data,
                                       const ObjectiveFunction*, int num_tree_per_iter) {
  if (cfg->data_sample_strategy == "goss") return new GOSSStrategy(cfg, data, num_tree_per_iter);
  return new BaggingStrategy(cfg, data);
}

// ------------------------------------------------------------------ GBDT
void GBDT::Init(const Config* config, const Dataset* train_data,
                const ObjectiveFunction* objective,
                const std::vector<const Metric*>& training_metrics) {
  config_store_ = *config;
  config_ = &config_store_;
  train_data_ = train_data;
  objective_ = objective;
  training_metrics_ = training_metrics;
  iter_ = 0;
  num_class_ = config_->num_class;
  num_tree_per_iteration_ = objective_ ? objective_->NumModelPerIteration() : num_class_;
  shrinkage_rate_ = config_->learning_rate;
  if (train_data_ != nullptr) {
    num_data_ = train_data_->num_data();
    max_feature_idx_ = train_data_->num_total_features() - 1;
    feature_names_ = train_data_->feature_names();
    // snapshot bin infos so model saving never needs the (possibly freed) dataset
    feature_infos_ = Common::SplitAny(train_data_->FeatureInfoString().c_str(), " ");
    label_idx_ = 0;
    tree_learner_.reset(TreeLearner::Create(config_->tree_learner, config_->device_type,
                                            config_));
    tree_learner_->Init(train_data_, GetIsConstHessian());
    train_score_.assign(static_cast<size_t>(num_data_) * num_tree_per_iteration_, 0.0);
    const double* init_sc = train_data_->metadata().init_score();
    if (init_sc != nullptr) {
      int64_t len = train_data_->metadata().num_init_score();
      std::copy(init_sc, init_sc + std::min<int64_t>(len, train_score_.size()),
                train_score_.begin());
      // device-resident scores must start from the same per-row init offsets
      if (tree_learner_->IsHIPLearner()) tree_learner_->UploadTrainScore(train_score_.data());
    }
    // continued training (input_model loaded before Init): fold the existing
    // trees into the scores so new gradients continue from them
    for (size_t i = 0; i < models_.size(); ++i) {
      const int tid = static_cast<int>(i) % num_tree_per_iteration_;
      models_[i]->AddPredictionToScoreByValue(
          train_data_, num_data_,
          train_score_.data() + static_cast<size_t>(tid) * num_data_);
    }
    if (!models_.empty() && tree_learner_->IsHIPLearner())
      tree_learner_->UploadTrainScore(train_score_.data());
    gradients_.assign(train_score_.size(), 0);
    hessians_.assign(train_score_.size(), 0);
    sample_strategy_.reset(SampleStrategy::Create(config_, train_data_, objective_,
                                                  num_tree_per_iteration_));
  }
  if (objective_) objective_name_ = objective_->GetName();
  init_scores_.assign(num_tree_per_iteration_, 0.0);
  best_metric_.clear();
}

void GBDT::ResetTrainingData(const Dataset* train_data, const ObjectiveFunction* objective,
                             const std::vector<const Metric*>& training_metrics) {
  train_data_ = train_data;
  objective_ = objective;
  training_metrics_ = training_metrics;
  num_data_ = train_data_->num_data();
  // a file-loaded booster has no learner/config until (re)initialized for training
  if (tree_learner_ != nullptr) tree_learner_->ResetTrainingData(train_data_);
  train_score_.assign(static_cast<size_t>(num_data_) * num_tree_per_iteration_, 0.0);
  // re-apply existing model to scores — ByValue walk: file-loaded models carry
  // only real-valued thresholds (also exact for in-memory trees)
  for (size_t i = 0; i < models_.size(); ++i) {
    int tid = static_cast<int>(i) % num_tree_per_iteration_;
    models_[i]->AddPredictionToScoreByValue(
        train_data_, num_data_,
        train_score_.data() + static_cast<size_t>(tid) * num_data_);
  }
  gradients_.assign(train_score_.size(), 0);
  hessians_.assign(train_score_.size(), 0);
  if (config_ != nullptr) {
    sample_strategy_.reset(SampleStrategy::Create(config_, train_data_, objective_,
                                                  num_tree_per_iteration_));
  }
}

void GBDT::ResetConfig(const Config* config) {
  config_store_ = *config;
  config_ = &config_store_;
  shrinkage_rate_ = config_->learning_rate;
  if (tree_learner_) tree_learner_->ResetConfig(config_);
  if (train_data_)
    sample_strategy_.reset(SampleStrategy::Create(config_, train_data_, objective_,
                                                  num_tree_per_iteration_));
}

void GBDT::AddValidDataset(const Dataset* valid_data,
                           const std::vector<const Metric*>& valid_metrics) {
  valid_data_.push_back(valid_data);
  valid_metrics_.push_back(valid_metrics);
  std::vector<double> score(static_cast<size_t>(valid_data->num_data()) *
                            num_tree_per_iteration_, 0.0);
  const double* init_sc = valid_data->metadata().init_score();
  if (init_sc != nullptr) {
    int64_t len = valid_data->metadata().num_init_score();
    std::copy(init_sc, init_sc + std::min<int64_t>(len, score.size()), score.begin());
  }
  // apply existing model (ByValue: merged/loaded trees carry no bin thresholds)
  for (size_t i = 0; i < models_.size(); ++i) {
    int tid = static_cast<int>(i) % num_tree_per_iteration_;
    models_[i]->AddPredictionToScoreByValue(
        valid_data, valid_data->num_data(),
        score.data() + static_cast<size_t>(tid) * valid_data->num_data());
  }
  valid_score_.push_back(std::move(score));
}

double GBDT::BoostFromAverage(int class_id, bool update_scores) {
  if (models_.empty() && !train_score_.empty() && objective_ != nullptr &&
      config_->boost_from_average && train_data_->metadata().init_score() == nullptr) {
    double init_score = objective_->BoostFromScore(class_id);
    init_score = Network::GlobalSyncUpByMean(init_score);
    if (std::fabs(init_score) > kEpsilon && update_scores) {
      if (tree_learner_->IsHIPLearner()) {
        tree_learner_->SetClassOffset(class_id);
        tree_learner_->DeviceAddInitScore(init_score);
      }
      double* sc = train_score_.data() + static_cast<size_t>(class_id) * num_data_;
#pragma omp parallel for schedule(static)
      for (data_size_t i = 0; i < num_data_; ++i) sc[i] += init_score;
      for (size_t v = 0; v < valid_data_.size(); ++v) {
        double* vs = valid_score_[v].data() +
                     static_cast<size_t>(class_id) * valid_data_[v]->num_data();
        data_size_t vn = valid_data_[v]->num_data();
        for (data_size_t i = 0; i < vn; ++i) vs[i] += init_score;
      }
    }
    return init_score;
  }
  return 0.0;
}

bool GBDT::TrainOneIter(const score_t* gradients, const score_t* hessians) {
  std::vector<double> init_scores(num_tree_per_iteration_, 0.0);
  const bool dev = tree_learner_->IsHIPLearner();
  // GOSS reads and rescales gradients on the host, so device-resident boosting
  // is disabled for gradient-dependent sample strategies (grads are computed on
  // the host and uploaded by the HIP learner instead)
  const bool dev_obj = dev && objective_ != nullptr &&
                       !sample_strategy_->NeedsGradients() &&
                       tree_learner_->DeviceObjectiveSupported(objective_->GetName());
  if (gradients == nullptr || hessians == nullptr) {
    for (int c = 0; c < num_tree_per_iteration_; ++c)
      init_scores[c] = BoostFromAverage(c, true);
    if (dev_obj) {
      tree_learner_->DeviceBoosting(objective_);
      gradients = gradients_.data();   // sentinel (ignored by the HIP learner)
      hessians = hessians_.data();
    } else {
      if (dev) tree_learner_->DownloadTrainScore(train_score_.data());
      Timer::Global().Start("boosting");
      objective_->GetGradients(train_score_.data(), gradients_.data(), hessians_.data());
      Timer::Global().Stop("boosting");
      gradients = gradients_.data();
      hessians = hessians_.data();
    }
  } else {
    // custom objective: copy so bagging/GOSS can modify
    std::copy(gradients, gradients + gradients_.size(), gradients_.begin());
    std::copy(hessians, hessians + hessians_.size(), hessians_.begin());
    gradients = gradients_.data();
    hessians = hessians_.data();
  }
  // bagging
  sample_strategy_->Bagging(iter_, tree_learner_.get(), gradients_.data(), hessians_.data());

  bool should_continue = false;
  for (int c = 0; c < num_tree_per_iteration_; ++c) {
    const size_t off = static_cast<size_t>(c) * num_data_;
    std::unique_ptr<Tree> new_tree(new Tree(2));
    tree_learner_->SetClassOffset(c);
    if (objective_ == nullptr || objective_->ClassNeedTrain(c)) {
      Timer::Global().Start("train_tree");
      new_tree.reset(tree_learner_->Train(gradients + off, hessians + off, models_.empty()));
      Timer::Global().Stop("train_tree");
    }
    if (new_tree->num_leaves() > 1) {
      should_continue = true;
      if (objective_ != nullptr && objective_->NeedRenewTreeOutput()) {
        tree_learner_->RenewTreeOutput(new_tree.get(), objective_, nullptr, num_data_,
                                       sample_strategy_->bag_indices().data(),
                                       sample_strategy_->bag_cnt(),
                                       train_score_.data() + off);
      }
      new_tree->Shrinkage(shrinkage_rate_);
      Timer::Global().Start("update_score");
      UpdateScore(new_tree.get(), c);
      Timer::Global().Stop("update_score");
      if (std::fabs(init_scores[c]) > kEpsilon) new_tree->AddBias(init_scores[c]);
    } else {
      // no splits: constant tree with objective-specific output
      if (models_.empty() && objective_ != nullptr) {
        double init_score = init_scores[c] != 0.0 ? init_scores[c]
                                                  : objective_->BoostFromScore(c);
        new_tree->AsConstantTree(init_score, num_data_);
        if (init_scores[c] == 0.0 && init_score != 0.0) {
          // apply to scores since BoostFromAverage didn't
          double* sc = train_score_.data() + off;
#pragma omp parallel for schedule(static)
          for (data_size_t i = 0; i < num_data_; ++i) sc[i] += init_score;
        }
      }
    }
    models_.push_back(std::move(new_tree));
  }
  if (!should_continue) {
    Log::Warning("Stopped training because there are no more leaves that meet the split "
                 "requirements");
    if (models_.size() > static_cast<size_t>(num_tree_per_iteration_)) {
      for (int c = 0; c < num_tree_per_iteration_; ++c) models_.pop_back();
    }
    return true;
  }
  ++iter_;
  return false;
}

void GBDT::RollbackOneIter() {
  if (iter_ <= 0) return;
  for (int c = 0; c < num_tree_per_iteration_; ++c) {
    auto& tree = models_[models_.size() - num_tree_per_iteration_ + c];
    // subtract prediction
    std::vector<double> neg;
    Tree copy_for_sub(*tree);
    copy_for_sub.Shrinkage(-1.0);
    copy_for_sub.AddPredictionToScore(train_data_, num_data_,
                                      train_score_.data() + static_cast<size_t>(c) * num_data_);
    for (size_t v = 0; v < valid_data_.size(); ++v) {
      copy_for_sub.AddPredictionToScore(
          valid_data_[v], valid_data_[v]->num_data(),
          valid_score_[v].data() + static_cast<size_t>(c) * valid_data_[v]->num_data());
    }
  }
  for (int c = 0; c < num_tree_per_iteration_; ++c) models_.pop_back();
  --iter_;
}

void GBDT::UpdateScore(const Tree* tree, int cur_tree_id) {
  const size_t off = static_cast<size_t>(cur_tree_id) * num_data_;
  // bag rows via the learner's partition (O(n) gather), out-of-bag rows via tree walk
  tree_learner_->AddPredictionToScore(tree, train_score_.data() + off);
  if (sample_strategy_->bag_cnt() > 0 &&
      sample_strategy_->bag_cnt() < num_data_) {
    // out-of-bag rows
    const auto& bag = sample_strategy_->bag_indices();
    std::vector<data_size_t> oob;
    oob.reserve(num_data_ - sample_strategy_->bag_cnt());
    size_t bi = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      if (bi < bag.size() && bag[bi] == i) ++bi;
      else oob.push_back(i);
    }
    tree->AddPredictionToScore(train_data_, oob.data(),
                               static_cast<data_size_t>(oob.size()),
                               train_score_.data() + off);
  }
  for (size_t v = 0; v < valid_data_.size(); ++v) {
    tree->AddPredictionToScore(
        valid_data_[v], valid_data_[v]->num_data(),
        valid_score_[v].data() + static_cast<size_t>(cur_tree_id) * valid_data_[v]->num_data());
  }
}

/*! objective output-transform id for the device metric kernel:
 *  0 identity, 1 exp, 2 sigmoid(param), 3 logistic, 4 log1p(exp), 5 signed-square;
 *  -1 = unknown (forces the host eval path). */
static int DeviceConvertKind(const ObjectiveFunction* obj, const Config* cfg,
                             double* param) {
  *param = 0.0;
  if (obj == nullptr) return -1;
  const std::string name = obj->GetName();
  static const char* kIdentity[] = {"regression", "regression_l1", "huber", "fair",
                                    "quantile",   "mape",          "lambdarank",
                                    "rank_xendcg"};
  for (const char* s : kIdentity)
    if (name == s) return cfg->reg_sqrt ? 5 : 0;
  if (name == "poisson" || name == "gamma" || name == "tweedie") return 1;
  if (name == "binary") {
    *param = cfg->sigmoid;
    return 2;
  }
  if (name == "cross_entropy") return 3;
  if (name == "cross_entropy_lambda") return 4;
  return -1;
}

std::vector<double> GBDT::GetEvalAt(int data_idx) const {
  std::vector<double> out;
  if (data_idx == 0 && tree_learner_ && tree_learner_->IsHIPLearner()) {
    // device fast path: pointwise train metrics reduce (Σ w·loss, Σ w) on the
    // GPU; the full score vector is downloaded only if some metric needs it
    // (AUC/NDCG/..). Capability parity: reference cuda_pointwise_metric.cu.
    double cparam = 0.0;
    const int ckind = DeviceConvertKind(objective_, config_, &cparam);
    std::vector<const Metric*> host_metrics;
    std::vector<std::pair<size_t, double>> dev_vals;  // (output slot, value)
    size_t slot = 0;
    bool all_dev = ckind >= 0;
    for (auto* m : training_metrics_) {
      const auto desc = m->pointwise_desc();
      double s = 0, w = 0;
      if (all_dev && desc.kind >= 0 &&
          tree_learner_->DeviceEvalPointwise(desc.kind, desc.a,
                                             desc.convert ? ckind : 0, cparam, &s, &w)) {
        dev_vals.emplace_back(slot, m->FinalizeFromSums(s, w));
        slot += 1;
      } else {
        host_metrics.push_back(m);
        slot += m->GetName().size();
      }
    }
    if (!host_metrics.empty()) {
      tree_learner_->DownloadTrainScore(const_cast<double*>(train_score_.data()));
    }
    // rebuild outputs in metric order
    size_t dev_i = 0;
    for (auto* m : training_metrics_) {
      const auto desc = m->pointwise_desc();
      if (all_dev && desc.kind >= 0 && dev_i < dev_vals.size() &&
          dev_vals[dev_i].first == out.size()) {
        out.push_back(dev_vals[dev_i].second);
        ++dev_i;
      } else {
        auto r = m->Eval(train_score_.data(), objective_);
        out.insert(out.end(), r.begin(), r.end());
      }
    }
    return out;
  }
  if (data_idx == 0) {
    for (auto* m : training_metrics_) {
      auto r = m->Eval(train_score_.data(), objective_);
      out.insert(out.end(), r.begin(), r.end());
    }
  } else {
    int v = data_idx - 1;
    for (auto* m : valid_metrics_[v]) {
      auto r = m->Eval(valid_score_[v].data(), objective_);
      out.insert(out.end(), r.begin(), r.end());
    }
  }
  return out;
}

std::vector<std::string> GBDT::EvalNames() const {
  std::vector<std::string> out;
  for (auto* m : training_metrics_) {
    for (auto& n : m->GetName()) out.push_back(n);
  }
  return out;
}

void GBDT::MergeFrom(const GBDT* other) {
  const int first_new = NumberOfTotalModel();
  for (int i = 0; i < other->NumberOfTotalModel(); ++i) {
    models_.emplace_back(new Tree(*other->models_[i]));
  }
  iter_ += other->iter_;
  // continued training must SEE the merged trees: fold their outputs into the
  // train/valid scores so the next iteration's gradients continue from them
  // (the reference reaches the same state by loading the model before Init)
  if (train_data_ != nullptr && !train_score_.empty()) {
    const bool dev = tree_learner_ != nullptr && tree_learner_->IsHIPLearner();
    for (int i = first_new; i < NumberOfTotalModel(); ++i) {
      const int cls = (i - first_new) % num_tree_per_iteration_;
      // ByValue walk: merged trees come from model text and carry only
      // real-valued thresholds (no bin-space thresholds)
      models_[i]->AddPredictionToScoreByValue(
          train_data_, num_data_,
          train_score_.data() + static_cast<size_t>(cls) * num_data_);
      for (size_t v = 0; v < valid_data_.size(); ++v) {
        models_[i]->AddPredictionToScoreByValue(
            valid_data_[v], valid_data_[v]->num_data(),
            valid_score_[v].data() +
                static_cast<size_t>(cls) * valid_data_[v]->num_data());
      }
    }
    if (dev) tree_learner_->UploadTrainScore(train_score_.data());
  }
}

bool GBDT::EvalAndCheckEarlyStopping() {
  if (config_->early_stopping_round <= 0 || valid_data_.empty()) return false;
  if (valid_metrics_.empty() || valid_metrics_[0].empty()) return false;
  // reference semantics: each (valid set, metric) pair tracks its own best; a
  // pair that fails to improve for early_stopping_round evals triggers the stop.
  // first_metric_only restricts tracking to the first metric of each valid set.
  size_t slot = 0;
  bool stop = false;
  for (size_t vd = 0; vd < valid_data_.size(); ++vd) {
    for (size_t mi = 0; mi < valid_metrics_[vd].size(); ++mi) {
      if (config_->first_metric_only && mi > 0) continue;
      const Metric* m = valid_metrics_[vd][mi];
      auto r = m->Eval(valid_score_[vd].data(), objective_);
      const double v = r[0] * m->factor_to_bigger_better();
      if (best_metric_.size() <= slot) {
        best_metric_.resize(slot + 1, std::numeric_limits<double>::infinity());
        es_counts_.resize(slot + 1, 0);
      }
      if (v < best_metric_[slot] - config_->early_stopping_min_delta) {
        best_metric_[slot] = v;
        es_counts_[slot] = 0;
        if (vd == 0 && mi == 0) best_iter_ = iter_;
      } else {
        if (++es_counts_[slot] >= config_->early_stopping_round) stop = true;
      }
      ++slot;
    }
  }
  if (stop)
    Log::Info("Early stopping at iteration %d, best iteration %d", iter_, best_iter_);
  return stop;
}

std::string GBDT::OutputMetric(int iter) {
  std::stringstream ss;
  for (size_t v = 0; v < valid_data_.size(); ++v) {
    for (auto* m : valid_metrics_[v]) {
      auto r = m->Eval(valid_score_[v].data(), objective_);
      for (size_t k = 0; k < r.size(); ++k)
        ss << "Iteration:" << iter << ", valid_" << (v + 1) << " " << m->GetName()[k] << " : "
           << r[k] << "\n";
    }
  }
  if (config_->is_provide_training_metric) {
    for (auto* m : training_metrics_) {
      auto r = m->Eval(train_score_.data(), objective_);
      for (size_t k = 0; k < r.size(); ++k)
        ss << "Iteration:" << iter << ", training " << m->GetName()[k] << " : " << r[k] << "\n";
    }
  }
  return ss.str();
}

void GBDT::Train(int snapshot_freq, const std::string& model_output_path) {
  for (int i = 0; i < config_->num_iterations; ++i) {
    if (TrainOneIter(nullptr, nullptr)) break;
    if (config_->metric_freq > 0 && (i + 1) % config_->metric_freq == 0) {
      auto s = OutputMetric(i + 1);
      if (!s.empty()) Log::Info("%s", s.c_str());
    }
    if (EvalAndCheckEarlyStopping()) break;
    if (snapshot_freq > 0 && (i + 1) % snapshot_freq == 0 && !model_output_path.empty()) {
      SaveModelToFile(0, -1, 0,
                      (model_output_path + ".snapshot_iter_" + std::to_string(i + 1)).c_str());
    }
  }
}

// ------------------------------------------------------------------ prediction
void GBDT::PredictRaw(const double* features, double* output, int start_iter,
                      int num_iter) const {
  const int total_iters = static_cast<int>(models_.size()) / num_tree_per_iteration_;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  for (int c = 0; c < num_tree_per_iteration_; ++c) output[c] = 0.0;
  for (int it = start_iter; it < end_iter; ++it) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) {
      output[c] += models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c]->Predict(features);
    }
  }
  if (average_output_ && end_iter > start_iter) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) output[c] /= (end_iter - start_iter);
  }
}

void GBDT::PredictRawEarlyStop(const double* features, double* output, int start_iter,
                               int num_iter, int round_period, double margin_threshold,
                               bool multiclass) const {
  const int total_iters = static_cast<int>(models_.size()) / num_tree_per_iteration_;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  for (int c = 0; c < num_tree_per_iteration_; ++c) output[c] = 0.0;
  int since_check = 0;
  for (int it = start_iter; it < end_iter; ++it) {
    for (int c = 0; c < num_tree_per_iteration_; ++c)
      output[c] += models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c]
                       ->Predict(features);
    if (++since_check >= round_period) {
      since_check = 0;
      double margin;
      if (multiclass) {
        double best = output[0], second = -1e308;
        for (int c = 1; c < num_tree_per_iteration_; ++c) {
          if (output[c] > best) { second = best; best = output[c]; }
          else if (output[c] > second) second = output[c];
        }
        margin = best - second;
      } else {
        margin = 2.0 * std::fabs(output[0]);
      }
      if (margin >= margin_threshold) break;
    }
  }
  if (average_output_ && end_iter > start_iter) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) output[c] /= (end_iter - start_iter);
  }
}

void GBDT::Predict(const double* features, double* output, int start_iter, int num_iter) const {
  PredictRaw(features, output, start_iter, num_iter);
  if (objective_ != nullptr) objective_->ConvertOutput(output, output);
}

void GBDT::PredictLeafIndex(const double* features, double* output, int start_iter,
                            int num_iter) const {
  const int total_iters = static_cast<int>(models_.size()) / num_tree_per_iteration_;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  int k = 0;
  for (int it = start_iter; it < end_iter; ++it) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) {
      output[k++] = models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c]
                        ->PredictLeafIndex(features);
    }
  }
}

void GBDT::PredictContrib(const double* features, double* output, int start_iter,
                          int num_iter) const {
  // SHAP values via per-tree path attribution (TreeSHAP); implemented in predict_contrib.cpp
  extern void TreeSHAP(const Tree* tree, const double* features, double* phi, int num_features);
  const int nf = max_feature_idx_ + 1;
  const int total_iters = static_cast<int>(models_.size()) / num_tree_per_iteration_;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  std::fill(output, output + static_cast<size_t>(num_tree_per_iteration_) * (nf + 1), 0.0);
  for (int it = start_iter; it < end_iter; ++it) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) {
      TreeSHAP(models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c].get(), features,
               output + static_cast<size_t>(c) * (nf + 1), nf);
    }
  }
}

const double* GBDT::GetTrainingScore(int64_t* out_len) const {
  if (tree_learner_ && tree_learner_->IsHIPLearner())
    tree_learner_->DownloadTrainScore(const_cast<double*>(train_score_.data()));
  *out_len = static_cast<int64_t>(train_score_.size());
  return train_score_.data();
}

int64_t GBDT::GetNumPredictAt(int data_idx) const {
  if (data_idx == 0) return static_cast<int64_t>(train_score_.size());
  return static_cast<int64_t>(valid_score_[data_idx - 1].size());
}

void GBDT::GetPredictAt(int data_idx, double* result, int64_t* out_len) const {
  if (data_idx == 0 && tree_learner_ && tree_learner_->IsHIPLearner())
    tree_learner_->DownloadTrainScore(const_cast<double*>(train_score_.data()));
  const std::vector<double>* sc =
      data_idx == 0 ? &train_score_ : &valid_score_[data_idx - 1];
  data_size_t n = data_idx == 0 ? num_data_ : valid_data_[data_idx - 1]->num_data();
  *out_len = static_cast<int64_t>(sc->size());
  if (objective_ != nullptr && num_tree_per_iteration_ == objective_->NumPredictOneRow()) {
    // convert per row
    if (num_tree_per_iteration_ == 1) {
      for (data_size_t i = 0; i < n; ++i) objective_->ConvertOutput(&(*sc)[i], &result[i]);
    } else {
      std::vector<double> in(num_tree_per_iteration_), out(num_tree_per_iteration_);
      for (data_size_t i = 0; i < n; ++i) {
        for (int c = 0; c < num_tree_per_iteration_; ++c)
          in[c] = (*sc)[static_cast<size_t>(c) * n + i];
        objective_->ConvertOutput(in.data(), out.data());
        for (int c = 0; c < num_tree_per_iteration_; ++c)
          result[static_cast<size_t>(c) * n + i] = out[c];
      }
    }
  } else {
    std::copy(sc->begin(), sc->end(), result);
  }
}

int GBDT::NumPredictOneRow(int start_iter, int num_iter, bool predict_leaf,
                           bool contrib) const {
  const int total_iters = static_cast<int>(models_.size()) / num_tree_per_iteration_;
  int end_iter = num_iter <= 0 ? total_iters : std::min(total_iters, start_iter + num_iter);
  int iters = std::max(0, end_iter - start_iter);
  if (predict_leaf) return iters * num_tree_per_iteration_;
  if (contrib) return num_tree_per_iteration_ * (max_feature_idx_ + 2);
  return num_tree_per_iteration_;
}

double GBDT::GetUpperBoundValue() const {
  double s = 0;
  for (auto& t : models_) s += std::max(0.0, t->GetUpperBoundValue());
  return s;
}
double GBDT::GetLowerBoundValue() const {
  double s = 0;
  for (auto& t : models_) s += std::min(0.0, t->GetLowerBoundValue());
  return s;
}

void GBDT::RefitTree(const int32_t* leaf_preds, int nrow, int ncol) {
  MIGBM_CHECK_EQ(static_cast<size_t>(ncol), models_.size());
  // re-derive gradients at current scores, refit leaf outputs with decay
  if (tree_learner_ && tree_learner_->IsHIPLearner())
    tree_learner_->DownloadTrainScore(train_score_.data());
  objective_->GetGradients(train_score_.data(), gradients_.data(), hessians_.data());
  for (int t = 0; t < ncol; ++t) {
    int tid = t % num_tree_per_iteration_;
    const size_t off = static_cast<size_t>(tid) * num_data_;
    Tree* tree = models_[t].get();
    const int nl = tree->num_leaves();
    std::vector<double> sum_g(nl, 0.0), sum_h(nl, 0.0);
    for (data_size_t i = 0; i < std::min<data_size_t>(nrow, num_data_); ++i) {
      int leaf = leaf_preds[static_cast<size_t>(i) * ncol + t];
      if (leaf < 0 || leaf >= nl) continue;
      sum_g[leaf] += gradients_[off + i];
      sum_h[leaf] += hessians_[off + i];
    }
    for (int l = 0; l < nl; ++l) {
      double new_out = GainMath::CalculateSplittedLeafOutput(
          sum_g[l], sum_h[l], config_->lambda_l1, config_->lambda_l2, config_->max_delta_step);
      double old = tree->LeafOutput(l);
      tree->SetLeafOutput(l, config_->refit_decay_rate * old +
                                 (1.0 - config_->refit_decay_rate) * new_out * shrinkage_rate_);
    }
  }
}

// ------------------------------------------------------------------ DART
std::vector<int> DART::DroppingTrees() {
  // parity: reference dart.hpp DroppingTrees — skip_drop gate, then per-tree
  // Bernoulli with weight-proportional rates (uniform_drop=false default) and a
  // max_drop-derived rate clamp; an empty selection is a plain boosting step.
  if (!drop_rng_init_) { drop_rng_ = Random(config_->drop_seed); drop_rng_init_ = true; }
  std::vector<int> dropped;
  const int n_iters = iter_;
  if (n_iters == 0) return dropped;
  if (drop_rng_.NextFloat() < config_->skip_drop) return dropped;
  double drop_rate = config_->drop_rate;
  if (!config_->uniform_drop && sum_weight_ > 0) {
    const double inv_avg = static_cast<double>(tree_weight_.size()) / sum_weight_;
    if (config_->max_drop > 0)
      drop_rate = std::min(drop_rate, config_->max_drop * inv_avg / sum_weight_);
    for (int i = 0; i < n_iters; ++i) {
      if (drop_rng_.NextFloat() < drop_rate * tree_weight_[i] * inv_avg) {
        dropped.push_back(i);
        if (static_cast<int>(dropped.size()) >= config_->max_drop) break;
      }
    }
  } else {
    if (config_->max_drop > 0)
      drop_rate = std::min(drop_rate, config_->max_drop / static_cast<double>(n_iters));
    for (int i = 0; i < n_iters; ++i) {
      if (drop_rng_.NextFloat() < drop_rate) {
        dropped.push_back(i);
        if (static_cast<int>(dropped.size()) >= config_->max_drop) break;
      }
    }
  }
  return dropped;
}

bool DART::TrainOneIter(const score_t* gradients, const score_t* hessians) {
  // continued training: merged/loaded iterations have no recorded drop weights —
  // pad them at weight 1.0 so absolute iteration indexing stays in bounds
  while (tree_weight_.size() < static_cast<size_t>(GetCurrentIteration())) {
    tree_weight_.push_back(1.0);
    sum_weight_ += 1.0;
  }
  auto dropped = DroppingTrees();
  // device-resident scores: DART's drop/renormalize surgery happens on the host
  // copy, synced down before and up after each host-side mutation
  const bool dev = tree_learner_->IsHIPLearner();
  if (dev) tree_learner_->DownloadTrainScore(train_score_.data());
  auto apply_tree_all = [&](const Tree& t, int c) {
    // add t's prediction to train and all valid score buffers for class c
    const_cast<Tree&>(t).AddPredictionToScore(
        train_data_, num_data_, train_score_.data() + static_cast<size_t>(c) * num_data_);
    for (size_t v = 0; v < valid_data_.size(); ++v) {
      const_cast<Tree&>(t).AddPredictionToScore(
          valid_data_[v], valid_data_[v]->num_data(),
          valid_score_[v].data() + static_cast<size_t>(c) * valid_data_[v]->num_data());
    }
  };
  // subtract dropped trees from scores
  for (int it : dropped) {
    for (int c = 0; c < num_tree_per_iteration_; ++c) {
      Tree neg(*models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c]);
      neg.Shrinkage(-1.0);
      apply_tree_all(neg, c);
    }
  }
  if (dev) tree_learner_->UploadTrainScore(train_score_.data());
  // the new tree's weight: lr/(1+k), or the xgboost-mode lr/(lr+k) (reference
  // dart.hpp shrinkage_rate_); applied directly through the base train path
  const double k = static_cast<double>(dropped.size());
  const double lr = config_->learning_rate;
  const double saved_rate = shrinkage_rate_;
  if (config_->xgboost_dart_mode)
    shrinkage_rate_ = dropped.empty() ? lr : lr / (lr + k);
  else
    shrinkage_rate_ = lr / (1.0 + k);
  const double new_weight = shrinkage_rate_;
  bool stop = GBDT::TrainOneIter(gradients, hessians);
  shrinkage_rate_ = saved_rate;
  if (!stop && dev) tree_learner_->DownloadTrainScore(train_score_.data());
  if (!stop) {
    tree_weight_.push_back(new_weight);
    sum_weight_ += new_weight;
    // dropped trees re-enter at factor k/(k+1) (or k/(k+lr) in xgboost mode)
    const double factor = config_->xgboost_dart_mode ? k / (k + lr) : k / (k + 1.0);
    for (int c = 0; c < num_tree_per_iteration_; ++c) {
      for (int it : dropped) {
        Tree* old = models_[static_cast<size_t>(it) * num_tree_per_iteration_ + c].get();
        old->Shrinkage(factor);
        apply_tree_all(*old, c);
      }
    }
    for (int it : dropped) {
      sum_weight_ -= tree_weight_[it] * (1.0 - factor);
      tree_weight_[it] *= factor;
    }
    if (dev) tree_learner_->UploadTrainScore(train_score_.data());
  }
  return stop;
}

// ------------------------------------------------------------------ RF
void RF::Init(const Config* config, const Dataset* train_data,
              const ObjectiveFunction* objective,
              const std::vector<const Metric*>& training_metrics) {
  GBDT::Init(config, train_data, objective, training_metrics);
  average_output_ = true;
  shrinkage_rate_ = 1.0;
  if (config->bagging_freq <= 0 || config->bagging_fraction >= 1.0)
    Log::Warning("RF requires bagging (bagging_freq>0 and bagging_fraction<1)");
}

bool RF::TrainOneIter(const score_t* gradients, const score_t* hessians) {
  // RF: gradients always from the ORIGINAL scores (zero), trees averaged
  std::fill(train_score_.begin(), train_score_.end(), 0.0);
  if (gradients == nullptr || hessians == nullptr) {
    objective_->GetGradients(train_score_.data(), gradients_.data(), hessians_.data());
  } else {
    std::copy(gradients, gradients + gradients_.size(), gradients_.begin());
    std::copy(hessians, hessians + hessians_.size(), hessians_.begin());
  }
  sample_strategy_->Bagging(iter_, tree_learner_.get(), gradients_.data(), hessians_.data());
  bool should_continue = false;
  for (int c = 0; c < num_tree_per_iteration_; ++c) {
    const size_t off = static_cast<size_t>(c) * num_data_;
    std::unique_ptr<Tree> new_tree(
        tree_learner_->Train(gradients_.data() + off, hessians_.data() + off, models_.empty()));
    if (new_tree->num_leaves() > 1) should_continue = true;
    models_.push_back(std::move(new_tree));
  }
  ++iter_;
  // keep valid scores as averaged raw sums (recomputed at eval time by GetEvalAt path)
  for (size_t v = 0; v < valid_data_.size(); ++v) {
    std::fill(valid_score_[v].begin(), valid_score_[v].end(), 0.0);
    for (size_t i = 0; i < models_.size(); ++i) {
      int tid = static_cast<int>(i) % num_tree_per_iteration_;
      models_[i]->AddPredictionToScore(
          valid_data_[v], valid_data_[v]->num_data(),
          valid_score_[v].data() + static_cast<size_t>(tid) * valid_data_[v]->num_data());
    }
    double inv = 1.0 / iter_;
    for (auto& s : valid_score_[v]) s *= inv;
  }
  return !should_continue;
}

}  // namespace migbm
