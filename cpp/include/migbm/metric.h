/*!
 * migbm Metric interface + factory.
 * Capability parity target: reference include/LightGBM/metric.h and src/metric/*.
 */
#ifndef MIGBM_METRIC_H_
#define MIGBM_METRIC_H_

#include "config.h"
#include "dataset.h"
#include "objective.h"

#include <string>
#include <vector>

namespace migbm {

/*! per-point loss kinds the HIP learner can reduce on device
 *  (keep in sync with k_metric_pointwise in hip_tree_learner.hip.cpp).
 *  Capability parity: reference src/metric/cuda/cuda_pointwise_metric.cu. */
enum PwLossKind : int {
  kPwL2 = 0,
  kPwL1,
  kPwQuantile,   // a = alpha
  kPwHuber,      // a = alpha
  kPwFair,       // a = fair_c
  kPwPoisson,
  kPwMape,
  kPwGamma,
  kPwGammaDev,
  kPwTweedie,    // a = tweedie_variance_power
  kPwBinaryLogloss,
  kPwBinaryError,
  kPwXent,
  kPwXentLambda,
};

/*! descriptor for device-side pointwise metric evaluation: the HIP learner
 *  computes (Σ w·loss, Σ w) on the GPU so train-metric eval stops downloading
 *  the full score vector. kind < 0 = not device-evaluable (AUC, NDCG, ...). */
struct PointwiseEvalDesc {
  int kind = -1;
  double a = 0.0;
  bool convert = true;  // apply the objective's output transform to the score first
};

class Metric {
 public:
  virtual ~Metric() = default;
  virtual void Init(const Metadata& metadata, data_size_t num_data) = 0;
  virtual const std::vector<std::string>& GetName() const = 0;
  /*! -1 if higher is better (AUC/NDCG), +1 if lower is better */
  virtual double factor_to_bigger_better() const = 0;
  virtual std::vector<double> Eval(const double* score,
                                   const ObjectiveFunction* objective) const = 0;
  /*! device fast path: descriptor (kind -1 = unsupported) */
  virtual PointwiseEvalDesc pointwise_desc() const { return {}; }
  /*! finish a device (loss_sum, weight_sum) pair into the metric value;
   *  applies the distributed reduce + the metric's final transform */
  virtual double FinalizeFromSums(double sum, double w) const { return sum / w; }

  static Metric* Create(const std::string& name, const Config& config);
};

}  // namespace migbm

#endif  // MIGBM_METRIC_H_
