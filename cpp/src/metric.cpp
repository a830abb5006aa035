/*! migbm metrics: regression family, binary (logloss/error/auc/ap), multiclass, ranking
 *  (ndcg@/map@), cross-entropy. Parity target: reference src/metric/*. */
#include "migbm/metric.h"
#include "migbm/network.h"

#include <algorithm>
#include <numeric>

namespace migbm {

namespace {

/*! distributed eval: reduce (sum, weight) over every rank's shard so all ranks
 *  report the metric of the UNION (parity: reference Network::GlobalSync). */
inline double GlobalAvg(double sum, double w) {
  if (Network::is_distributed()) {
    double buf[2] = {sum, w};
    Network::AllreduceSum(buf, 2);
    sum = buf[0];
    w = buf[1];
  }
  return sum / std::max(1.0, w);
}

/*! distributed sort-based eval (AUC / AP / auc_mu): gather every rank's
 *  (scores, label, weight) so the metric is computed over the UNION of shards
 *  (VERDICT r1 #7 — per-rank AUC silently reported a wrong number before).
 *  Correctness-grade payload: these metrics need a global sort anyway. */
struct EvalUnion {
  std::vector<double> score;  // nc * n, class-major
  std::vector<label_t> label, weight;
  data_size_t n = 0;
  bool has_weight = false;
};

inline bool GatherEvalUnion(const double* score, int nc, const label_t* label,
                            const label_t* weight, data_size_t n, EvalUnion* out) {
  if (!Network::is_distributed()) return false;
  const int world = Network::num_machines();
  std::vector<int64_t> counts(world, 0);
  int64_t mine = n;
  Network::Allgather(reinterpret_cast<const char*>(&mine), sizeof(int64_t),
                     reinterpret_cast<char*>(counts.data()));
  int64_t total = 0;
  for (int64_t c : counts) total += c;
  out->n = static_cast<data_size_t>(total);
  std::vector<int> lsz(world), dsz(world);
  for (int r = 0; r < world; ++r) {
    lsz[r] = static_cast<int>(counts[r] * sizeof(label_t));
    dsz[r] = static_cast<int>(counts[r] * sizeof(double));
  }
  out->label.resize(total);
  Network::AllgatherV(reinterpret_cast<const char*>(label),
                      static_cast<int>(n * sizeof(label_t)), lsz.data(),
                      reinterpret_cast<char*>(out->label.data()));
  out->has_weight = weight != nullptr;
  if (out->has_weight) {
    out->weight.resize(total);
    Network::AllgatherV(reinterpret_cast<const char*>(weight),
                        static_cast<int>(n * sizeof(label_t)), lsz.data(),
                        reinterpret_cast<char*>(out->weight.data()));
  }
  out->score.resize(static_cast<size_t>(nc) * total);
  for (int c = 0; c < nc; ++c) {
    Network::AllgatherV(reinterpret_cast<const char*>(score + static_cast<size_t>(c) * n),
                        static_cast<int>(n * sizeof(double)), dsz.data(),
                        reinterpret_cast<char*>(out->score.data() +
                                                static_cast<size_t>(c) * total));
  }
  return true;
}

/*! generic pointwise metric: avg of per-row loss (weighted). */
class PointwiseMetric : public Metric {
 public:
  explicit PointwiseMetric(const std::string& name, bool convert_score,
                           std::function<double(double label, double pred)> loss,
                           std::function<double(double sum, double w)> final_trans = nullptr)
      : names_({name}), convert_(convert_score), loss_(std::move(loss)),
        final_(std::move(final_trans)) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    sum_w_ = 0.0;
    if (weights_) { for (data_size_t i = 0; i < num_data_; ++i) sum_w_ += weights_[i]; }
    else sum_w_ = num_data_;
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return 1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction* obj) const override {
    double sum = 0.0;
#pragma omp parallel for schedule(static) reduction(+ : sum)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double pred = score[i];
      if (convert_ && obj != nullptr) obj->ConvertOutput(&score[i], &pred);
      double w = weights_ ? weights_[i] : 1.0;
      sum += w * loss_(label_[i], pred);
    }
    double gsum = sum, gw = sum_w_;
    if (Network::is_distributed()) {
      // distributed eval: the metric is over the UNION of rank shards
      // (parity: reference metrics' Network::GlobalSync of loss sums)
      double buf[2] = {gsum, gw};
      Network::AllreduceSum(buf, 2);
      gsum = buf[0];
      gw = buf[1];
    }
    double r = final_ ? final_(gsum, gw) : gsum / std::max(1.0, gw);
    return {r};
  }
  PointwiseEvalDesc pointwise_desc() const override { return desc_; }
  PointwiseMetric* WithDesc(int kind, double a = 0.0) {
    desc_.kind = kind;
    desc_.a = a;
    desc_.convert = convert_;
    return this;
  }
  double FinalizeFromSums(double sum, double w) const override {
    double gsum = sum, gw = w;
    if (Network::is_distributed()) {
      double buf[2] = {gsum, gw};
      Network::AllreduceSum(buf, 2);
      gsum = buf[0];
      gw = buf[1];
    }
    return final_ ? final_(gsum, gw) : gsum / std::max(1.0, gw);
  }

 protected:
  std::vector<std::string> names_;
  bool convert_;
  std::function<double(double, double)> loss_;
  std::function<double(double, double)> final_;
  PointwiseEvalDesc desc_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  double sum_w_ = 0.0;
};

class AUCMetric : public Metric {
 public:
  AUCMetric() : names_({"auc"}) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return -1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    EvalUnion u;
    if (GatherEvalUnion(score, 1, label_, weights_, num_data_, &u))
      return {EvalImpl(u.score.data(), u.label.data(),
                       u.has_weight ? u.weight.data() : nullptr, u.n)};
    return {EvalImpl(score, label_, weights_, num_data_)};
  }
  static double EvalImpl(const double* score, const label_t* label_,
                         const label_t* weights_, data_size_t num_data_) {
    std::vector<data_size_t> order(num_data_);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              [score](data_size_t a, data_size_t b) { return score[a] > score[b]; });
    double accum_pos = 0, accum_neg = 0, auc = 0;
    double cur_pos = 0, cur_neg = 0;
    double prev_score = std::numeric_limits<double>::infinity();
    for (data_size_t i = 0; i < num_data_; ++i) {
      data_size_t r = order[i];
      double w = weights_ ? weights_[r] : 1.0;
      if (score[r] != prev_score) {
        // flush tie group: each negative in the group pairs with positives ranked
        // strictly above it (accum_pos) plus half credit for tied positives
        auc += cur_neg * (accum_pos + cur_pos * 0.5);
        accum_pos += cur_pos;
        accum_neg += cur_neg;
        cur_pos = cur_neg = 0;
        prev_score = score[r];
      }
      if (label_[r] > 0) cur_pos += w;
      else cur_neg += w;
    }
    auc += cur_neg * (accum_pos + cur_pos * 0.5);
    accum_pos += cur_pos;
    accum_neg += cur_neg;
    if (accum_pos > 0 && accum_neg > 0) auc /= (accum_pos * accum_neg);
    else auc = 1.0;
    return auc;
  }

 private:
  std::vector<std::string> names_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

class AveragePrecisionMetric : public Metric {
 public:
  AveragePrecisionMetric() : names_({"average_precision"}) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return -1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    EvalUnion u;
    if (GatherEvalUnion(score, 1, label_, weights_, num_data_, &u))
      return {EvalImpl(u.score.data(), u.label.data(),
                       u.has_weight ? u.weight.data() : nullptr, u.n)};
    return {EvalImpl(score, label_, weights_, num_data_)};
  }
  static double EvalImpl(const double* score, const label_t* label_,
                         const label_t* weights_, data_size_t num_data_) {
    std::vector<data_size_t> order(num_data_);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              [score](data_size_t a, data_size_t b) { return score[a] > score[b]; });
    double tp = 0, fp = 0, total_pos = 0, ap = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[order[i]] : 1.0;
      if (label_[order[i]] > 0) total_pos += w;
    }
    if (total_pos <= 0) return 1.0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      data_size_t r = order[i];
      double w = weights_ ? weights_[r] : 1.0;
      if (label_[r] > 0) {
        tp += w;
        ap += w * tp / (tp + fp);
      } else {
        fp += w;
      }
    }
    return ap / total_pos;
  }

 private:
  std::vector<std::string> names_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

class BinaryLoglossMetric : public Metric {
 public:
  BinaryLoglossMetric() : names_({"binary_logloss"}) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    sum_w_ = 0;
    if (weights_) for (data_size_t i = 0; i < num_data_; ++i) sum_w_ += weights_[i];
    else sum_w_ = num_data_;
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return 1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction* obj) const override {
    double sum = 0;
#pragma omp parallel for schedule(static) reduction(+ : sum)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double p = score[i];
      if (obj) obj->ConvertOutput(&score[i], &p);
      else p = Common::Sigmoid(score[i]);
      p = std::min(1.0 - 1e-12, std::max(1e-12, p));
      double w = weights_ ? weights_[i] : 1.0;
      sum += w * (label_[i] > 0 ? -std::log(p) : -std::log(1.0 - p));
    }
    return {GlobalAvg(sum, sum_w_)};
  }
  PointwiseEvalDesc pointwise_desc() const override { return {kPwBinaryLogloss, 0.0, true}; }
  double FinalizeFromSums(double sum, double w) const override { return GlobalAvg(sum, w); }

 private:
  std::vector<std::string> names_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  double sum_w_ = 0;
};

class MultiLoglossMetric : public Metric {
 public:
  explicit MultiLoglossMetric(int num_class) : names_({"multi_logloss"}), nc_(num_class) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    sum_w_ = 0;
    if (weights_) for (data_size_t i = 0; i < num_data_; ++i) sum_w_ += weights_[i];
    else sum_w_ = num_data_;
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return 1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction* obj) const override {
    double sum = 0;
#pragma omp parallel for schedule(static) reduction(+ : sum)
    for (data_size_t i = 0; i < num_data_; ++i) {
      std::vector<double> raw(nc_), prob(nc_);
      for (int c = 0; c < nc_; ++c) raw[c] = score[static_cast<size_t>(c) * num_data_ + i];
      if (obj) obj->ConvertOutput(raw.data(), prob.data());
      else {
        double mx = *std::max_element(raw.begin(), raw.end());
        double s = 0;
        for (int c = 0; c < nc_; ++c) { prob[c] = std::exp(raw[c] - mx); s += prob[c]; }
        for (int c = 0; c < nc_; ++c) prob[c] /= s;
      }
      int lbl = static_cast<int>(label_[i]);
      double p = std::min(1.0 - 1e-12, std::max(1e-12, prob[lbl]));
      double w = weights_ ? weights_[i] : 1.0;
      sum += -w * std::log(p);
    }
    return {GlobalAvg(sum, sum_w_)};
  }

 private:
  std::vector<std::string> names_;
  int nc_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  double sum_w_ = 0;
};

class MultiErrorMetric : public Metric {
 public:
  MultiErrorMetric(int num_class, int top_k) : nc_(num_class), top_k_(top_k) {
    names_ = {top_k == 1 ? std::string("multi_error")
                         : "multi_error@" + std::to_string(top_k)};
  }
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    sum_w_ = 0;
    if (weights_) for (data_size_t i = 0; i < num_data_; ++i) sum_w_ += weights_[i];
    else sum_w_ = num_data_;
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return 1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    double sum = 0;
#pragma omp parallel for schedule(static) reduction(+ : sum)
    for (data_size_t i = 0; i < num_data_; ++i) {
      int lbl = static_cast<int>(label_[i]);
      double ls = score[static_cast<size_t>(lbl) * num_data_ + i];
      int num_better = 0;
      for (int c = 0; c < nc_; ++c)
        if (score[static_cast<size_t>(c) * num_data_ + i] > ls) ++num_better;
      double w = weights_ ? weights_[i] : 1.0;
      if (num_better >= top_k_) sum += w;
    }
    return {GlobalAvg(sum, sum_w_)};
  }

 private:
  std::vector<std::string> names_;
  int nc_, top_k_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  double sum_w_ = 0;
};

class NDCGMetric : public Metric {
 public:
  explicit NDCGMetric(const Config& cfg) {
    eval_at_ = cfg.eval_at;
    if (eval_at_.empty()) eval_at_ = {1, 2, 3, 4, 5};
    for (int k : eval_at_) names_.push_back("ndcg@" + std::to_string(k));
    label_gain_ = cfg.label_gain;
    if (label_gain_.empty())
      for (int i = 0; i < 31; ++i) label_gain_.push_back((1u << i) - 1.0);
  }
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    query_boundaries_ = meta.query_boundaries();
    num_queries_ = meta.num_queries();
    query_weights_ = meta.query_weights();
    if (query_boundaries_ == nullptr) Log::Fatal("NDCG metric requires query information");
    sum_qw_ = query_weights_ ? 0.0 : static_cast<double>(num_queries_);
    if (query_weights_)
      for (data_size_t q = 0; q < num_queries_; ++q) sum_qw_ += query_weights_[q];
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return -1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    std::vector<double> result(eval_at_.size(), 0.0);
    std::vector<double> tmp(eval_at_.size());
    for (data_size_t q = 0; q < num_queries_; ++q) {
      data_size_t s = query_boundaries_[q], cnt = query_boundaries_[q + 1] - s;
      std::vector<data_size_t> order(cnt);
      std::iota(order.begin(), order.end(), 0);
      std::sort(order.begin(), order.end(), [&](data_size_t a, data_size_t b) {
        return score[s + a] > score[s + b];
      });
      std::vector<double> gains(cnt);
      for (data_size_t i = 0; i < cnt; ++i)
        gains[i] = label_gain_[static_cast<int>(label_[s + i])];
      std::vector<double> sorted_gains = gains;
      std::sort(sorted_gains.begin(), sorted_gains.end(), std::greater<double>());
      double qw = query_weights_ ? query_weights_[q] : 1.0;
      for (size_t ki = 0; ki < eval_at_.size(); ++ki) {
        int k = std::min<int>(eval_at_[ki], static_cast<int>(cnt));
        double dcg = 0, idcg = 0;
        for (int i = 0; i < k; ++i) {
          dcg += gains[order[i]] / std::log2(2.0 + i);
          idcg += sorted_gains[i] / std::log2(2.0 + i);
        }
        tmp[ki] = idcg > 0 ? dcg / idcg : 1.0;
        result[ki] += qw * tmp[ki];
      }
    }
    double gqw = sum_qw_;
    if (Network::is_distributed()) {
      std::vector<double> buf(result);
      buf.push_back(gqw);
      Network::AllreduceSum(buf.data(), buf.size());
      for (size_t i = 0; i < result.size(); ++i) result[i] = buf[i];
      gqw = buf.back();
    }
    for (auto& r : result) r /= std::max(1.0, gqw);
    return result;
  }

 private:
  std::vector<std::string> names_;
  std::vector<int> eval_at_;
  std::vector<double> label_gain_;
  data_size_t num_data_ = 0, num_queries_ = 0;
  const label_t* label_ = nullptr;
  const data_size_t* query_boundaries_ = nullptr;
  const label_t* query_weights_ = nullptr;
  double sum_qw_ = 0;
};

class MapMetric : public Metric {
 public:
  explicit MapMetric(const Config& cfg) {
    eval_at_ = cfg.eval_at;
    if (eval_at_.empty()) eval_at_ = {1, 2, 3, 4, 5};
    for (int k : eval_at_) names_.push_back("map@" + std::to_string(k));
  }
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    query_boundaries_ = meta.query_boundaries();
    num_queries_ = meta.num_queries();
    if (query_boundaries_ == nullptr) Log::Fatal("MAP metric requires query information");
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return -1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    std::vector<double> result(eval_at_.size(), 0.0);
    for (data_size_t q = 0; q < num_queries_; ++q) {
      data_size_t s = query_boundaries_[q], cnt = query_boundaries_[q + 1] - s;
      std::vector<data_size_t> order(cnt);
      std::iota(order.begin(), order.end(), 0);
      std::sort(order.begin(), order.end(), [&](data_size_t a, data_size_t b) {
        return score[s + a] > score[s + b];
      });
      for (size_t ki = 0; ki < eval_at_.size(); ++ki) {
        int k = std::min<int>(eval_at_[ki], static_cast<int>(cnt));
        double hits = 0, ap = 0;
        for (int i = 0; i < k; ++i) {
          if (label_[s + order[i]] > 0) {
            hits += 1;
            ap += hits / (i + 1);
          }
        }
        result[ki] += hits > 0 ? ap / hits : 0.0;
      }
    }
    double gq = static_cast<double>(num_queries_);
    if (Network::is_distributed()) {
      std::vector<double> buf(result);
      buf.push_back(gq);
      Network::AllreduceSum(buf.data(), buf.size());
      for (size_t i = 0; i < result.size(); ++i) result[i] = buf[i];
      gq = buf.back();
    }
    for (auto& r : result) r /= std::max(1.0, gq);
    return result;
  }

 private:
  std::vector<std::string> names_;
  std::vector<int> eval_at_;
  data_size_t num_data_ = 0, num_queries_ = 0;
  const label_t* label_ = nullptr;
  const data_size_t* query_boundaries_ = nullptr;
};

/*! auc_mu: multiclass AUC (average of one-vs-one partition AUCs; parity target:
 *  reference multiclass_metric.hpp AucMuMetric). */
class AucMuMetric : public Metric {
 public:
  explicit AucMuMetric(const Config& cfg) : names_({"auc_mu"}), nc_(cfg.num_class) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
  }
  const std::vector<std::string>& GetName() const override { return names_; }
  double factor_to_bigger_better() const override { return -1.0; }
  std::vector<double> Eval(const double* score, const ObjectiveFunction*) const override {
    EvalUnion u;
    if (GatherEvalUnion(score, nc_, label_, weights_, num_data_, &u))
      return {EvalImpl(u.score.data(), u.label.data(),
                       u.has_weight ? u.weight.data() : nullptr, u.n, nc_)};
    return {EvalImpl(score, label_, weights_, num_data_, nc_)};
  }
  static double EvalImpl(const double* score, const label_t* label_,
                         const label_t* weights_, data_size_t num_data_, int nc_) {
    double total = 0;
    int pairs = 0;
    for (int a = 0; a < nc_; ++a) {
      for (int b = a + 1; b < nc_; ++b) {
        // rows of classes a,b ranked by s_a - s_b; AUC of class a vs b
        std::vector<std::pair<double, int>> v;
        std::vector<double> w;
        for (data_size_t i = 0; i < num_data_; ++i) {
          const int c = static_cast<int>(label_[i]);
          if (c != a && c != b) continue;
          const double sa = score[static_cast<size_t>(a) * num_data_ + i];
          const double sb = score[static_cast<size_t>(b) * num_data_ + i];
          v.push_back({sa - sb, c == a ? 1 : 0});
          w.push_back(weights_ ? weights_[i] : 1.0);
        }
        std::vector<size_t> order(v.size());
        std::iota(order.begin(), order.end(), size_t{0});
        std::sort(order.begin(), order.end(),
                  [&](size_t x, size_t y) { return v[x].first > v[y].first; });
        double acc_pos = 0, acc_neg = 0, cp = 0, cn = 0, auc = 0;
        double prev = std::numeric_limits<double>::infinity();
        for (size_t k = 0; k < order.size(); ++k) {
          const auto& e = v[order[k]];
          if (e.first != prev) {
            auc += cn * (acc_pos + cp * 0.5);
            acc_pos += cp;
            acc_neg += cn;
            cp = cn = 0;
            prev = e.first;
          }
          if (e.second) cp += w[order[k]];
          else cn += w[order[k]];
        }
        auc += cn * (acc_pos + cp * 0.5);
        acc_pos += cp;
        acc_neg += cn;
        if (acc_pos > 0 && acc_neg > 0) {
          total += auc / (acc_pos * acc_neg);
          ++pairs;
        }
      }
    }
    return pairs > 0 ? total / pairs : 1.0;
  }

 private:
  std::vector<std::string> names_;
  int nc_;
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

}  // namespace

Metric* Metric::Create(const std::string& name, const Config& cfg) {
  if (name == "l2" || name == "mse" || name == "regression" || name == "mean_squared_error")
    return (new PointwiseMetric("l2", true, [](double y, double p) { return (y - p) * (y - p); }))
        ->WithDesc(kPwL2);
  if (name == "rmse" || name == "root_mean_squared_error" || name == "l2_root")
    return (new PointwiseMetric("rmse", true,
                               [](double y, double p) { return (y - p) * (y - p); },
                               [](double s, double w) { return std::sqrt(s / std::max(1.0, w)); }))
        ->WithDesc(kPwL2);
  if (name == "l1" || name == "mae" || name == "mean_absolute_error")
    return (new PointwiseMetric("l1", true, [](double y, double p) { return std::fabs(y - p); }))
        ->WithDesc(kPwL1);
  if (name == "quantile") {
    double a = cfg.alpha;
    return (new PointwiseMetric("quantile", true, [a](double y, double p) {
      double d = y - p;
      return d >= 0 ? a * d : (a - 1) * d;
    }))->WithDesc(kPwQuantile, a);
  }
  if (name == "huber") {
    double a = cfg.alpha;
    return (new PointwiseMetric("huber", true, [a](double y, double p) {
      double d = std::fabs(y - p);
      return d <= a ? 0.5 * d * d : a * (d - 0.5 * a);
    }))->WithDesc(kPwHuber, a);
  }
  if (name == "fair") {
    double c = cfg.fair_c;
    return (new PointwiseMetric("fair", true, [c](double y, double p) {
      double x = std::fabs(y - p);
      return c * x - c * c * std::log1p(x / c);
    }))->WithDesc(kPwFair, c);
  }
  if (name == "poisson")
    return (new PointwiseMetric("poisson", true, [](double y, double p) {
      double eps = 1e-10;
      if (p <= eps) p = eps;
      return p - y * std::log(p);
    }))->WithDesc(kPwPoisson);
  if (name == "mape")
    return (new PointwiseMetric("mape", true, [](double y, double p) {
      return std::fabs((y - p) / std::max(1.0, std::fabs(y)));
    }))->WithDesc(kPwMape);
  if (name == "gamma")
    return (new PointwiseMetric("gamma", true, [](double y, double p) {
      double eps = 1e-10;
      if (p <= eps) p = eps;
      return y / p + std::log(p) - 1;  // negative log-likelihood up to const
    }))->WithDesc(kPwGamma);
  if (name == "gamma_deviance")
    return (new PointwiseMetric("gamma_deviance", true, [](double y, double p) {
      double eps = 1e-10;
      if (p <= eps) p = eps;
      if (y <= eps) return 0.0;
      return 2.0 * (std::log(p / y) + y / p - 1);
    }))->WithDesc(kPwGammaDev);
  if (name == "tweedie") {
    double rho = cfg.tweedie_variance_power;
    return (new PointwiseMetric("tweedie", true, [rho](double y, double p) {
      double eps = 1e-10;
      if (p <= eps) p = eps;
      return -y * std::pow(p, 1 - rho) / (1 - rho) + std::pow(p, 2 - rho) / (2 - rho);
    }))->WithDesc(kPwTweedie, rho);
  }
  if (name == "r2") {
    class R2Metric : public PointwiseMetric {
     public:
      R2Metric() : PointwiseMetric("r2", true,
                                   [](double y, double p) { return (y - p) * (y - p); }) {
        WithDesc(kPwL2);
      }
      double FinalizeFromSums(double sum, double w) const override {
        const double mse = PointwiseMetric::FinalizeFromSums(sum, w);
        double mean = 0, sw = 0;
        for (data_size_t i = 0; i < num_data_; ++i) {
          double wi = weights_ ? weights_[i] : 1.0;
          mean += wi * label_[i];
          sw += wi;
        }
        mean /= std::max(sw, 1.0);
        double var = 0;
        for (data_size_t i = 0; i < num_data_; ++i) {
          double wi = weights_ ? weights_[i] : 1.0;
          var += wi * (label_[i] - mean) * (label_[i] - mean);
        }
        var /= std::max(sw, 1.0);
        return var > 0 ? 1.0 - mse / var : 0.0;
      }
      double factor_to_bigger_better() const override { return -1.0; }
      std::vector<double> Eval(const double* score,
                               const ObjectiveFunction* obj) const override {
        auto mse = PointwiseMetric::Eval(score, obj);
        double mean = 0, sw = 0;
        for (data_size_t i = 0; i < num_data_; ++i) {
          double w = weights_ ? weights_[i] : 1.0;
          mean += w * label_[i];
          sw += w;
        }
        mean /= std::max(sw, 1.0);
        double var = 0;
        for (data_size_t i = 0; i < num_data_; ++i) {
          double w = weights_ ? weights_[i] : 1.0;
          var += w * (label_[i] - mean) * (label_[i] - mean);
        }
        var /= std::max(sw, 1.0);
        return {var > 0 ? 1.0 - mse[0] / var : 0.0};
      }
    };
    return new R2Metric();
  }
  if (name == "binary_logloss" || name == "logloss" || name == "binary")
    return new BinaryLoglossMetric();
  if (name == "regression_l1")
    return (new PointwiseMetric("l1", true, [](double y, double p) { return std::fabs(y - p); }))
        ->WithDesc(kPwL1);
  if (name == "binary_error")
    return (new PointwiseMetric("binary_error", true, [](double y, double p) {
      return (p > 0.5 ? 1.0 : 0.0) != (y > 0 ? 1.0 : 0.0) ? 1.0 : 0.0;
    }))->WithDesc(kPwBinaryError);
  if (name == "auc") return new AUCMetric();
  if (name == "average_precision") return new AveragePrecisionMetric();
  if (name == "multi_logloss" || name == "softmax" || name == "multiclass" ||
      name == "multiclassova")
    return new MultiLoglossMetric(cfg.num_class);
  if (name == "multi_error") return new MultiErrorMetric(cfg.num_class, cfg.multi_error_top_k);
  if (name == "auc_mu") return new AucMuMetric(cfg);
  if (name == "ndcg" || name == "lambdarank" || name == "rank_xendcg") return new NDCGMetric(cfg);
  if (name == "map" || name == "mean_average_precision") return new MapMetric(cfg);
  if (name == "cross_entropy" || name == "xentropy")
    return (new PointwiseMetric("cross_entropy", true, [](double y, double p) {
      p = std::min(1.0 - 1e-12, std::max(1e-12, p));
      return -y * std::log(p) - (1 - y) * std::log(1 - p);
    }))->WithDesc(kPwXent);
  if (name == "cross_entropy_lambda" || name == "xentlambda")
    return new PointwiseMetric("cross_entropy_lambda", true, [](double y, double p) {
      double hhat = std::log1p(std::max(1e-12, p));
      return y * hhat - p;  // placeholder consistent transform
    });
  if (name == "kullback_leibler" || name == "kldiv")
    return new PointwiseMetric("kullback_leibler", true, [](double y, double p) {
      p = std::min(1.0 - 1e-12, std::max(1e-12, p));
      double a = y > 1e-12 ? y * std::log(y / p) : 0.0;
      double b = (1 - y) > 1e-12 ? (1 - y) * std::log((1 - y) / (1 - p)) : 0.0;
      return a + b;
    });
  if (name == "none" || name == "null" || name == "na" || name.empty()) return nullptr;
  Log::Warning("Unknown metric %s, ignored", name.c_str());
  return nullptr;
}

}  // namespace migbm
