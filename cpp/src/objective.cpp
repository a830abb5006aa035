/*! migbm objective functions: regression family, binary, multiclass, cross-entropy,
 *  lambdarank, rank_xendcg. Gradient formulas per the published algorithms (parity target:
 *  reference src/objective/*). */
#include "migbm/objective.h"

#include <algorithm>
#include <numeric>

namespace migbm {

namespace {

/*! weighted percentile of values (alpha in [0,1]); matches reference PercentileFun. */
template <typename Getter>
double Percentile(double alpha, data_size_t cnt, Getter val) {
  if (cnt == 0) return 0.0;
  std::vector<double> v(cnt);
  for (data_size_t i = 0; i < cnt; ++i) v[i] = val(i);
  std::sort(v.begin(), v.end());
  double pos = alpha * (cnt - 1);
  data_size_t lo = static_cast<data_size_t>(pos);
  data_size_t hi = std::min<data_size_t>(lo + 1, cnt - 1);
  double frac = pos - lo;
  return v[lo] * (1 - frac) + v[hi] * frac;
}

template <typename Getter, typename WGetter>
double WeightedPercentile(double alpha, data_size_t cnt, Getter val, WGetter wgt) {
  if (cnt == 0) return 0.0;
  std::vector<data_size_t> order(cnt);
  std::iota(order.begin(), order.end(), 0);
  std::sort(order.begin(), order.end(),
            [&](data_size_t a, data_size_t b) { return val(a) < val(b); });
  double total = 0;
  for (data_size_t i = 0; i < cnt; ++i) total += wgt(i);
  double target = alpha * total, acc = 0;
  for (data_size_t i = 0; i < cnt; ++i) {
    acc += wgt(order[i]);
    if (acc >= target) return val(order[i]);
  }
  return val(order[cnt - 1]);
}

}  // namespace

// ------------------------------------------------------------------ regression family
class RegressionL2loss : public ObjectiveFunction {
 public:
  explicit RegressionL2loss(const Config& cfg) : sqrt_(cfg.reg_sqrt) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    if (sqrt_) {
      trans_label_.resize(num_data_);
      for (data_size_t i = 0; i < num_data_; ++i) {
        double l = label_[i];
        trans_label_[i] = static_cast<label_t>(l >= 0 ? std::sqrt(l) : -std::sqrt(-l));
      }
      label_ = trans_label_.data();
    }
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      grad[i] = static_cast<score_t>(w * (score[i] - label_[i]));
      hess[i] = static_cast<score_t>(w);
    }
  }
  bool IsConstantHessian() const override { return weights_ == nullptr; }
  double BoostFromScore(int) const override {
    double s = 0, w = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      double wi = weights_ ? weights_[i] : 1.0;
      s += label_[i] * wi;
      w += wi;
    }
    return w > 0 ? s / w : 0.0;
  }
  void ConvertOutput(const double* in, double* out) const override {
    if (sqrt_) { double v = *in; *out = v * std::fabs(v); }
    else *out = *in;
  }
  const char* GetName() const override { return "regression"; }
  std::string ToString() const override {
    return sqrt_ ? "regression sqrt" : "regression";
  }

 protected:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  bool sqrt_ = false;
  std::vector<label_t> trans_label_;
};

class RegressionL1loss : public RegressionL2loss {
 public:
  explicit RegressionL1loss(const Config& cfg) : RegressionL2loss(cfg) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double d = score[i] - label_[i];
      grad[i] = static_cast<score_t>(d > 0 ? w : -w);
      hess[i] = static_cast<score_t>(w);
    }
  }
  bool IsConstantHessian() const override { return weights_ == nullptr; }
  double BoostFromScore(int) const override {
    if (weights_) {
      return WeightedPercentile(0.5, num_data_, [&](data_size_t i) { return (double)label_[i]; },
                                [&](data_size_t i) { return (double)weights_[i]; });
    }
    return Percentile(0.5, num_data_, [&](data_size_t i) { return (double)label_[i]; });
  }
  bool NeedRenewTreeOutput() const override { return true; }
  double RenewTreeOutput(double, const data_size_t* idx, data_size_t cnt,
                         const double* score) const override {
    if (weights_) {
      return WeightedPercentile(0.5, cnt,
                                [&](data_size_t i) { return label_[idx[i]] - score[idx[i]]; },
                                [&](data_size_t i) { return (double)weights_[idx[i]]; });
    }
    return Percentile(0.5, cnt,
                      [&](data_size_t i) { return label_[idx[i]] - score[idx[i]]; });
  }
  const char* GetName() const override { return "regression_l1"; }
  std::string ToString() const override { return "regression_l1"; }
};

class RegressionHuberLoss : public RegressionL2loss {
 public:
  explicit RegressionHuberLoss(const Config& cfg) : RegressionL2loss(cfg), alpha_(cfg.alpha) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double d = score[i] - label_[i];
      if (std::fabs(d) <= alpha_) grad[i] = static_cast<score_t>(w * d);
      else grad[i] = static_cast<score_t>(w * (d > 0 ? alpha_ : -alpha_));
      hess[i] = static_cast<score_t>(w);
    }
  }
  bool IsConstantHessian() const override { return weights_ == nullptr; }
  const char* GetName() const override { return "huber"; }

 private:
  double alpha_;
};

class RegressionFairLoss : public RegressionL2loss {
 public:
  explicit RegressionFairLoss(const Config& cfg) : RegressionL2loss(cfg), c_(cfg.fair_c) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double d = score[i] - label_[i];
      grad[i] = static_cast<score_t>(w * c_ * d / (std::fabs(d) + c_));
      hess[i] = static_cast<score_t>(w * c_ * c_ /
                                     ((std::fabs(d) + c_) * (std::fabs(d) + c_)));
    }
  }
  const char* GetName() const override { return "fair"; }

 private:
  double c_;
};

class RegressionPoissonLoss : public RegressionL2loss {
 public:
  explicit RegressionPoissonLoss(const Config& cfg)
      : RegressionL2loss(cfg), max_delta_step_(cfg.poisson_max_delta_step) {}
  void Init(const Metadata& meta, data_size_t n) override {
    RegressionL2loss::Init(meta, n);
    for (data_size_t i = 0; i < n; ++i)
      if (label_[i] < 0) Log::Fatal("Poisson objective requires non-negative labels");
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double e = std::exp(score[i]);
      grad[i] = static_cast<score_t>(w * (e - label_[i]));
      hess[i] = static_cast<score_t>(w * std::exp(score[i] + max_delta_step_));
    }
  }
  double BoostFromScore(int) const override {
    return std::log(std::max(kEpsilon, RegressionL2loss::BoostFromScore(0)));
  }
  void ConvertOutput(const double* in, double* out) const override { *out = std::exp(*in); }
  const char* GetName() const override { return "poisson"; }

 private:
  double max_delta_step_;
};

class RegressionQuantileloss : public RegressionL2loss {
 public:
  explicit RegressionQuantileloss(const Config& cfg)
      : RegressionL2loss(cfg), alpha_(cfg.alpha) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double d = score[i] - label_[i];
      grad[i] = static_cast<score_t>(d >= 0 ? w * (1.0 - alpha_) : -w * alpha_);
      hess[i] = static_cast<score_t>(w);
    }
  }
  bool IsConstantHessian() const override { return weights_ == nullptr; }
  double BoostFromScore(int) const override {
    if (weights_) {
      return WeightedPercentile(alpha_, num_data_,
                                [&](data_size_t i) { return (double)label_[i]; },
                                [&](data_size_t i) { return (double)weights_[i]; });
    }
    return Percentile(alpha_, num_data_, [&](data_size_t i) { return (double)label_[i]; });
  }
  bool NeedRenewTreeOutput() const override { return true; }
  double RenewTreeOutput(double, const data_size_t* idx, data_size_t cnt,
                         const double* score) const override {
    if (weights_) {
      return WeightedPercentile(alpha_, cnt,
                                [&](data_size_t i) { return label_[idx[i]] - score[idx[i]]; },
                                [&](data_size_t i) { return (double)weights_[idx[i]]; });
    }
    return Percentile(alpha_, cnt,
                      [&](data_size_t i) { return label_[idx[i]] - score[idx[i]]; });
  }
  const char* GetName() const override { return "quantile"; }
  std::string ToString() const override {
    return std::string("quantile alpha:") + Common::DoubleToStr(alpha_);
  }

 private:
  double alpha_;
};

class RegressionMAPELoss : public RegressionL1loss {
 public:
  explicit RegressionMAPELoss(const Config& cfg) : RegressionL1loss(cfg) {}
  void Init(const Metadata& meta, data_size_t n) override {
    RegressionL1loss::Init(meta, n);
    label_weights_.resize(n);
    for (data_size_t i = 0; i < n; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      label_weights_[i] = static_cast<label_t>(w / std::max(1.0, std::fabs((double)label_[i])));
    }
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double d = score[i] - label_[i];
      grad[i] = static_cast<score_t>(d > 0 ? label_weights_[i] : -label_weights_[i]);
      hess[i] = label_weights_[i];
    }
  }
  bool IsConstantHessian() const override { return false; }
  double BoostFromScore(int) const override {
    return WeightedPercentile(0.5, num_data_, [&](data_size_t i) { return (double)label_[i]; },
                              [&](data_size_t i) { return (double)label_weights_[i]; });
  }
  double RenewTreeOutput(double, const data_size_t* idx, data_size_t cnt,
                         const double* score) const override {
    return WeightedPercentile(0.5, cnt,
                              [&](data_size_t i) { return label_[idx[i]] - score[idx[i]]; },
                              [&](data_size_t i) { return (double)label_weights_[idx[i]]; });
  }
  const char* GetName() const override { return "mape"; }
  std::string ToString() const override { return "mape"; }

 private:
  std::vector<label_t> label_weights_;
};

class RegressionGammaLoss : public RegressionPoissonLoss {
 public:
  explicit RegressionGammaLoss(const Config& cfg) : RegressionPoissonLoss(cfg) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double e = std::exp(-score[i]);
      grad[i] = static_cast<score_t>(w * (1.0 - label_[i] * e));
      hess[i] = static_cast<score_t>(w * label_[i] * e);
    }
  }
  const char* GetName() const override { return "gamma"; }
};

class RegressionTweedieLoss : public RegressionPoissonLoss {
 public:
  explicit RegressionTweedieLoss(const Config& cfg)
      : RegressionPoissonLoss(cfg), rho_(cfg.tweedie_variance_power) {}
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double e1 = std::exp((1 - rho_) * score[i]);
      double e2 = std::exp((2 - rho_) * score[i]);
      grad[i] = static_cast<score_t>(w * (-label_[i] * e1 + e2));
      hess[i] = static_cast<score_t>(
          w * (-label_[i] * (1 - rho_) * e1 + (2 - rho_) * e2));
    }
  }
  const char* GetName() const override { return "tweedie"; }

 private:
  double rho_;
};

// ------------------------------------------------------------------ binary
class BinaryLogloss : public ObjectiveFunction {
 public:
  explicit BinaryLogloss(const Config& cfg)
      : sigmoid_(cfg.sigmoid), is_unbalance_(cfg.is_unbalance),
        scale_pos_weight_(cfg.scale_pos_weight) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    data_size_t pos = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      if (label_[i] != 0 && label_[i] != 1)
        Log::Fatal("Binary objective requires 0/1 labels (got %f)", (double)label_[i]);
      pos += label_[i] > 0 ? 1 : 0;
    }
    num_pos_ = pos;
    if (is_unbalance_ && pos > 0 && pos < num_data_) {
      data_size_t neg = num_data_ - pos;
      label_weight_pos_ = pos > neg ? 1.0 : static_cast<double>(neg) / pos;
      label_weight_neg_ = pos > neg ? static_cast<double>(pos) / neg : 1.0;
    } else {
      label_weight_pos_ = scale_pos_weight_;
      label_weight_neg_ = 1.0;
    }
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      const double y = label_[i] > 0 ? 1.0 : -1.0;
      const double lw = (y > 0 ? label_weight_pos_ : label_weight_neg_) *
                        (weights_ ? weights_[i] : 1.0);
      const double response = -y * sigmoid_ / (1.0 + std::exp(y * sigmoid_ * score[i]));
      const double abs_resp = std::fabs(response);
      grad[i] = static_cast<score_t>(response * lw);
      hess[i] = static_cast<score_t>(abs_resp * (sigmoid_ - abs_resp) * lw);
    }
  }
  double BoostFromScore(int) const override {
    double s = 0, w = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      double wi = weights_ ? weights_[i] : 1.0;
      s += (label_[i] > 0 ? 1.0 : 0.0) * wi;
      w += wi;
    }
    double p = std::min(1.0 - kEpsilon, std::max(kEpsilon, w > 0 ? s / w : 0.5));
    double init = std::log(p / (1.0 - p)) / sigmoid_;
    Log::Info("[binary] boost from score %f (pavg=%f)", init, p);
    return init;
  }
  bool ClassNeedTrain(int) const override { return num_pos_ > 0 && num_pos_ < num_data_; }
  void ConvertOutput(const double* in, double* out) const override {
    *out = 1.0 / (1.0 + std::exp(-sigmoid_ * (*in)));
  }
  data_size_t NumPositiveData() const override { return num_pos_; }
  const char* GetName() const override { return "binary"; }
  std::string ToString() const override {
    return std::string("binary sigmoid:") + Common::DoubleToStr(sigmoid_);
  }

 protected:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  double sigmoid_;
  bool is_unbalance_;
  double scale_pos_weight_;
  double label_weight_pos_ = 1.0, label_weight_neg_ = 1.0;
  data_size_t num_pos_ = 0;
};

// ------------------------------------------------------------------ multiclass
class MulticlassSoftmax : public ObjectiveFunction {
 public:
  explicit MulticlassSoftmax(const Config& cfg) : num_class_(cfg.num_class) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    class_counts_.assign(num_class_, 0);
    for (data_size_t i = 0; i < num_data_; ++i) {
      int c = static_cast<int>(label_[i]);
      if (c < 0 || c >= num_class_)
        Log::Fatal("Label %d out of range for num_class=%d", c, num_class_);
      class_counts_[c] += 1;
    }
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
    const double factor = static_cast<double>(num_class_) / (num_class_ - 1);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      // softmax over classes (class-major score layout)
      double mx = -1e300;
      for (int c = 0; c < num_class_; ++c)
        mx = std::max(mx, score[static_cast<size_t>(c) * num_data_ + i]);
      double sum = 0;
      for (int c = 0; c < num_class_; ++c)
        sum += std::exp(score[static_cast<size_t>(c) * num_data_ + i] - mx);
      const int lbl = static_cast<int>(label_[i]);
      for (int c = 0; c < num_class_; ++c) {
        double p = std::exp(score[static_cast<size_t>(c) * num_data_ + i] - mx) / sum;
        size_t k = static_cast<size_t>(c) * num_data_ + i;
        grad[k] = static_cast<score_t>(w * (p - (c == lbl ? 1.0 : 0.0)));
        hess[k] = static_cast<score_t>(w * factor * p * (1.0 - p));
      }
    }
  }
  double BoostFromScore(int class_id) const override {
    double p = std::max(kEpsilon, static_cast<double>(class_counts_[class_id]) /
                                       std::max<data_size_t>(1, num_data_));
    return std::log(p);
  }
  int NumModelPerIteration() const override { return num_class_; }
  int NumPredictOneRow() const override { return num_class_; }
  void ConvertOutput(const double* in, double* out) const override {
    double mx = *std::max_element(in, in + num_class_);
    double sum = 0;
    for (int c = 0; c < num_class_; ++c) { out[c] = std::exp(in[c] - mx); sum += out[c]; }
    for (int c = 0; c < num_class_; ++c) out[c] /= sum;
  }
  const char* GetName() const override { return "multiclass"; }
  std::string ToString() const override {
    return std::string("multiclass num_class:") + std::to_string(num_class_);
  }

 private:
  data_size_t num_data_ = 0;
  int num_class_;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  std::vector<data_size_t> class_counts_;
};

class MulticlassOVA : public ObjectiveFunction {
 public:
  explicit MulticlassOVA(const Config& cfg) : num_class_(cfg.num_class), sigmoid_(cfg.sigmoid) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      for (int c = 0; c < num_class_; ++c) {
        const double y = static_cast<int>(label_[i]) == c ? 1.0 : -1.0;
        size_t k = static_cast<size_t>(c) * num_data_ + i;
        const double response = -y * sigmoid_ / (1.0 + std::exp(y * sigmoid_ * score[k]));
        const double ar = std::fabs(response);
        grad[k] = static_cast<score_t>(response * w);
        hess[k] = static_cast<score_t>(ar * (sigmoid_ - ar) * w);
      }
    }
  }
  int NumModelPerIteration() const override { return num_class_; }
  int NumPredictOneRow() const override { return num_class_; }
  void ConvertOutput(const double* in, double* out) const override {
    for (int c = 0; c < num_class_; ++c) out[c] = 1.0 / (1.0 + std::exp(-sigmoid_ * in[c]));
  }
  const char* GetName() const override { return "multiclassova"; }
  std::string ToString() const override {
    return std::string("multiclassova num_class:") + std::to_string(num_class_) +
           " sigmoid:" + Common::DoubleToStr(sigmoid_);
  }

 private:
  data_size_t num_data_ = 0;
  int num_class_;
  double sigmoid_;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

// ------------------------------------------------------------------ cross entropy
class CrossEntropy : public ObjectiveFunction {
 public:
  explicit CrossEntropy(const Config&) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    for (data_size_t i = 0; i < num_data_; ++i)
      if (label_[i] < 0 || label_[i] > 1)
        Log::Fatal("cross_entropy requires labels in [0,1]");
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      double w = weights_ ? weights_[i] : 1.0;
      double p = 1.0 / (1.0 + std::exp(-score[i]));
      grad[i] = static_cast<score_t>(w * (p - label_[i]));
      hess[i] = static_cast<score_t>(w * p * (1.0 - p));
    }
  }
  double BoostFromScore(int) const override {
    double s = 0, w = 0;
    for (data_size_t i = 0; i < num_data_; ++i) {
      double wi = weights_ ? weights_[i] : 1.0;
      s += label_[i] * wi;
      w += wi;
    }
    double p = std::min(1.0 - kEpsilon, std::max(kEpsilon, s / std::max(w, kEpsilon)));
    return std::log(p / (1.0 - p));
  }
  void ConvertOutput(const double* in, double* out) const override {
    *out = 1.0 / (1.0 + std::exp(-*in));
  }
  const char* GetName() const override { return "cross_entropy"; }

 private:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

class CrossEntropyLambda : public ObjectiveFunction {
 public:
  explicit CrossEntropyLambda(const Config&) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      const double w = weights_ ? weights_[i] : 1.0;
      const double epf = std::exp(score[i]);
      const double hhat = std::log1p(epf);
      const double z = 1.0 - std::exp(-w * hhat);
      const double enf = 1.0 / epf;
      grad[i] = static_cast<score_t>((1.0 - label_[i] / z) * w / (1.0 + enf));
      const double c = 1.0 / (1.0 - z);
      double d = 1.0 + epf;
      const double a = w * epf / (d * d);
      hess[i] = static_cast<score_t>(
          a * (1.0 + label_[i] * (1.0 - c * (1.0 + w * epf / d * (1.0 - c)))));
    }
  }
  void ConvertOutput(const double* in, double* out) const override {
    *out = std::log1p(std::exp(*in));
  }
  const char* GetName() const override { return "cross_entropy_lambda"; }

 private:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
};

// ------------------------------------------------------------------ ranking
class LambdarankNDCG : public ObjectiveFunction {
 public:
  explicit LambdarankNDCG(const Config& cfg)
      : sigmoid_(cfg.sigmoid), norm_(cfg.lambdarank_norm),
        bias_reg_(cfg.lambdarank_position_bias_regularization),
        truncation_level_(cfg.lambdarank_truncation_level), label_gain_(cfg.label_gain) {
    if (label_gain_.empty()) {
      // default 2^i - 1
      for (int i = 0; i < 31; ++i) label_gain_.push_back((1u << i) - 1.0);
    }
  }
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    weights_ = meta.weights();
    query_boundaries_ = meta.query_boundaries();
    num_queries_ = meta.num_queries();
    positions_ = meta.positions();
    if (positions_ != nullptr) {
      int max_pos = 0;
      for (data_size_t i = 0; i < num_data_; ++i)
        max_pos = std::max(max_pos, positions_[i]);
      position_bias_.assign(max_pos + 1, 1.0);
      bias_num_.assign(max_pos + 1, 0.0);
      bias_den_.assign(max_pos + 1, 0.0);
      Log::Info("Lambdarank position debiasing enabled (%d positions)", max_pos + 1);
    }
    if (query_boundaries_ == nullptr)
      Log::Fatal("Lambdarank requires query information (group)");
    // inverse max DCG per query
    inverse_max_dcg_.resize(num_queries_);
#pragma omp parallel for schedule(static)
    for (data_size_t q = 0; q < num_queries_; ++q) {
      data_size_t s = query_boundaries_[q], e = query_boundaries_[q + 1];
      std::vector<double> gains;
      for (data_size_t i = s; i < e; ++i)
        gains.push_back(label_gain_[static_cast<int>(label_[i])]);
      std::sort(gains.begin(), gains.end(), std::greater<double>());
      double dcg = 0;
      int k = std::min<int>(truncation_level_, static_cast<int>(gains.size()));
      for (int i = 0; i < k; ++i) dcg += gains[i] / std::log2(2.0 + i);
      inverse_max_dcg_[q] = dcg > 0 ? 1.0 / dcg : 0.0;
    }
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
    if (positions_ != nullptr) {
      std::fill(bias_num_.begin(), bias_num_.end(), 0.0);
      std::fill(bias_den_.begin(), bias_den_.end(), 0.0);
    }
#pragma omp parallel for schedule(guided)
    for (data_size_t q = 0; q < num_queries_; ++q) {
      GetGradientsForOneQuery(q, score, grad, hess);
    }
    if (positions_ != nullptr) UpdatePositionBiasFactors();
  }

  /*! unbiased lambdarank: re-estimate per-position click propensity from the
   *  accumulated lambda mass (parity: reference rank_objective.hpp position
   *  debiasing — re-derived update rule). */
  void UpdatePositionBiasFactors() const {
    double base = bias_den_[0] > 0 ? bias_num_[0] / bias_den_[0] : 1.0;
    if (base <= 0) return;
    for (size_t k = 0; k < position_bias_.size(); ++k) {
      if (bias_den_[k] > 0) {
        double est = (bias_num_[k] / bias_den_[k]) / base;
        // L2 regularization pulls the propensity estimate toward 1 (analogue of
        // the reference's Newton-step regularizer, rank_objective.hpp:333-334)
        if (bias_reg_ > 0.0)
          est = 1.0 + (est - 1.0) / (1.0 + bias_reg_ * bias_den_[k]);
        // smoothed multiplicative update, clamped for stability
        position_bias_[k] = std::min(10.0, std::max(0.1, 0.9 * position_bias_[k] +
                                                              0.1 * est));
      }
    }
  }
  void GetGradientsForOneQuery(data_size_t q, const double* score_all, score_t* grad_all,
                               score_t* hess_all) const {
    const data_size_t s = query_boundaries_[q];
    const data_size_t cnt = query_boundaries_[q + 1] - s;
    const double* score = score_all + s;
    score_t* grad = grad_all + s;
    score_t* hess = hess_all + s;
    const label_t* label = label_ + s;
    for (data_size_t i = 0; i < cnt; ++i) { grad[i] = 0; hess[i] = 0; }
    if (inverse_max_dcg_[q] <= 0) return;
    // sorted indices by score desc
    std::vector<data_size_t> order(cnt);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              [score](data_size_t a, data_size_t b) { return score[a] > score[b]; });
    // best/worst score for delta clamp
    double best = score[order[0]];
    data_size_t worst_idx = cnt - 1;
    if (worst_idx > 0 && score[order[worst_idx]] == kMinScore) --worst_idx;
    const double worst = score[order[worst_idx]];
    double sum_lambdas = 0.0;
    const int trunc = std::min<int>(truncation_level_, static_cast<int>(cnt));
    for (int i = 0; i < trunc; ++i) {
      if (score[order[i]] == kMinScore) continue;
      for (data_size_t j = i + 1; j < cnt; ++j) {
        if (score[order[j]] == kMinScore) continue;
        if (label[order[i]] == label[order[j]]) continue;
        data_size_t high, low;
        int high_rank, low_rank;
        if (label[order[i]] > label[order[j]]) {
          high = order[i]; high_rank = i;
          low = order[j]; low_rank = static_cast<int>(j);
        } else {
          high = order[j]; high_rank = static_cast<int>(j);
          low = order[i]; low_rank = i;
        }
        const double high_label = label[high];
        const double low_label = label[low];
        const double delta_score = score[high] - score[low];
        const double high_gain = label_gain_[static_cast<int>(high_label)];
        const double low_gain = label_gain_[static_cast<int>(low_label)];
        const double high_disc = 1.0 / std::log2(2.0 + high_rank);
        const double low_disc = 1.0 / std::log2(2.0 + low_rank);
        double delta_pair_ndcg = (high_gain - low_gain) * (high_disc - low_disc) *
                                 inverse_max_dcg_[q];
        if (delta_pair_ndcg < 0) delta_pair_ndcg = -delta_pair_ndcg;
        if (norm_ && best != worst && high_label != low_label)
          delta_pair_ndcg /= (0.01 + std::fabs(delta_score));
        // lambda
        double p_lambda = 1.0 / (1.0 + std::exp(sigmoid_ * delta_score));
        double p_hessian = p_lambda * (1.0 - p_lambda);
        p_lambda *= -sigmoid_ * delta_pair_ndcg;
        p_hessian *= sigmoid_ * sigmoid_ * delta_pair_ndcg;
        if (positions_ != nullptr) {
          // inverse-propensity weighting by the displayed positions
          const int ph = positions_[s + high];
          const int pl = positions_[s + low];
          const double w_ipw = 1.0 / (position_bias_[ph] * position_bias_[pl]);
          p_lambda *= w_ipw;
          p_hessian *= w_ipw;
#pragma omp atomic
          bias_num_[ph] += std::fabs(p_lambda);
#pragma omp atomic
          bias_den_[ph] += 1.0;
#pragma omp atomic
          bias_num_[pl] += std::fabs(p_lambda);
#pragma omp atomic
          bias_den_[pl] += 1.0;
        }
        grad[high] += static_cast<score_t>(p_lambda);
        hess[high] += static_cast<score_t>(p_hessian);
        grad[low] -= static_cast<score_t>(p_lambda);
        hess[low] += static_cast<score_t>(p_hessian);
        sum_lambdas -= 2 * p_lambda;
      }
    }
    if (norm_ && sum_lambdas > 0) {
      double norm_factor = std::log2(1 + sum_lambdas) / sum_lambdas;
      for (data_size_t i = 0; i < cnt; ++i) {
        grad[i] = static_cast<score_t>(grad[i] * norm_factor);
        hess[i] = static_cast<score_t>(hess[i] * norm_factor);
      }
    }
    if (weights_) {
      for (data_size_t i = 0; i < cnt; ++i) {
        grad[i] = static_cast<score_t>(grad[i] * weights_[s + i]);
        hess[i] = static_cast<score_t>(hess[i] * weights_[s + i]);
      }
    }
  }
  const char* GetName() const override { return "lambdarank"; }

 protected:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const label_t* weights_ = nullptr;
  const data_size_t* query_boundaries_ = nullptr;
  data_size_t num_queries_ = 0;
  double sigmoid_;
  bool norm_;
  double bias_reg_ = 0.0;
  int truncation_level_;
  std::vector<double> label_gain_;
  std::vector<double> inverse_max_dcg_;
  const int32_t* positions_ = nullptr;
  mutable std::vector<double> position_bias_, bias_num_, bias_den_;
};

class RankXENDCG : public ObjectiveFunction {
 public:
  explicit RankXENDCG(const Config& cfg) : seed_(cfg.objective_seed) {}
  void Init(const Metadata& meta, data_size_t num_data) override {
    num_data_ = num_data;
    label_ = meta.label();
    query_boundaries_ = meta.query_boundaries();
    num_queries_ = meta.num_queries();
    if (query_boundaries_ == nullptr) Log::Fatal("rank_xendcg requires query information");
    rands_.clear();
    for (data_size_t i = 0; i < num_queries_; ++i) rands_.emplace_back(seed_ + i);
  }
  void GetGradients(const double* score, score_t* grad, score_t* hess) const override {
#pragma omp parallel for schedule(guided)
    for (data_size_t q = 0; q < num_queries_; ++q) {
      const data_size_t s = query_boundaries_[q];
      const data_size_t cnt = query_boundaries_[q + 1] - s;
      if (cnt == 1) { grad[s] = 0; hess[s] = 0; continue; }
      const double* sc = score + s;
      std::vector<double> rho(cnt);
      double mx = *std::max_element(sc, sc + cnt);
      double sum = 0;
      for (data_size_t i = 0; i < cnt; ++i) { rho[i] = std::exp(sc[i] - mx); sum += rho[i]; }
      for (data_size_t i = 0; i < cnt; ++i) rho[i] /= sum;
      // stochastic ground-truth distribution: phi_i = 2^l_i - u, u ~ U(0,1)
      // (XE-NDCG-MART; reference rank_objective.hpp:400-456 math)
      std::vector<double> t(cnt);
      Random& r = rands_[q];
      double denom = 0;
      for (data_size_t i = 0; i < cnt; ++i) {
        t[i] = std::pow(2.0, static_cast<int>(label_[s + i])) -
               static_cast<double>(r.NextFloat());
        denom += t[i];
      }
      const double inv_denom = 1.0 / std::max(1e-15, denom);
      // first-order term, then two softmax-weighted correction orders
      double sum_l1 = 0.0;
      for (data_size_t i = 0; i < cnt; ++i) {
        const double term = rho[i] - t[i] * inv_denom;
        grad[s + i] = static_cast<score_t>(term);
        t[i] = term / (1.0 - rho[i]);
        sum_l1 += t[i];
      }
      double sum_l2 = 0.0;
      for (data_size_t i = 0; i < cnt; ++i) {
        const double term = rho[i] * (sum_l1 - t[i]);
        grad[s + i] += static_cast<score_t>(term);
        t[i] = term / (1.0 - rho[i]);
        sum_l2 += t[i];
      }
      for (data_size_t i = 0; i < cnt; ++i) {
        grad[s + i] += static_cast<score_t>(rho[i] * (sum_l2 - t[i]));
        hess[s + i] = static_cast<score_t>(rho[i] * (1.0 - rho[i]));
      }
    }
  }
  const char* GetName() const override { return "rank_xendcg"; }

 private:
  data_size_t num_data_ = 0;
  const label_t* label_ = nullptr;
  const data_size_t* query_boundaries_ = nullptr;
  data_size_t num_queries_ = 0;
  int seed_;
  mutable std::vector<Random> rands_;
};

// ------------------------------------------------------------------ factory
ObjectiveFunction* ObjectiveFunction::Create(const std::string& name, const Config& cfg) {
  if (name == "regression") return new RegressionL2loss(cfg);
  if (name == "regression_l1") return new RegressionL1loss(cfg);
  if (name == "huber") return new RegressionHuberLoss(cfg);
  if (name == "fair") return new RegressionFairLoss(cfg);
  if (name == "poisson") return new RegressionPoissonLoss(cfg);
  if (name == "quantile") return new RegressionQuantileloss(cfg);
  if (name == "mape") return new RegressionMAPELoss(cfg);
  if (name == "gamma") return new RegressionGammaLoss(cfg);
  if (name == "tweedie") return new RegressionTweedieLoss(cfg);
  if (name == "binary") return new BinaryLogloss(cfg);
  if (name == "multiclass") return new MulticlassSoftmax(cfg);
  if (name == "multiclassova") return new MulticlassOVA(cfg);
  if (name == "cross_entropy") return new CrossEntropy(cfg);
  if (name == "cross_entropy_lambda") return new CrossEntropyLambda(cfg);
  if (name == "lambdarank") return new LambdarankNDCG(cfg);
  if (name == "rank_xendcg") return new RankXENDCG(cfg);
  if (name == "none" || name == "null" || name == "custom" || name == "na" || name.empty())
    return nullptr;
  Log::Fatal("Unknown objective: %s", name.c_str());
  return nullptr;
}

ObjectiveFunction* ObjectiveFunction::CreateFromModelString(const std::string& str) {
  auto toks = Common::SplitAny(str.c_str(), " ");
  if (toks.empty()) return nullptr;
  Config cfg;
  std::unordered_map<std::string, std::string> params;
  for (size_t i = 1; i < toks.size(); ++i) {
    auto kv = Common::Split(toks[i].c_str(), ':');
    if (kv.size() == 2) params[kv[0]] = kv[1];
  }
  cfg.Set(params);
  return Create(toks[0], cfg);
}

}  // namespace migbm
