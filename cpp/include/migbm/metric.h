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

class Metric {
 public:
  virtual ~Metric() = default;
  virtual void Init(const Metadata& metadata, data_size_t num_data) = 0;
  virtual const std::vector<std::string>& GetName() const = 0;
  /*! -1 if higher is better (AUC/NDCG), +1 if lower is better */
  virtual double factor_to_bigger_better() const = 0;
  virtual std::vector<double> Eval(const double* score,
                                   const ObjectiveFunction* objective) const = 0;

  static Metric* Create(const std::string& name, const Config& config);
};

}  // namespace migbm

#endif  // MIGBM_METRIC_H_
