/*!
 * migbm ObjectiveFunction interface + factory.
 * Capability parity target: reference include/LightGBM/objective_function.h and
 * src/objective/{regression,binary,multiclass,rank,xentropy}_objective.hpp. Fresh
 * implementation of the published gradient formulas.
 */
#ifndef MIGBM_OBJECTIVE_H_
#define MIGBM_OBJECTIVE_H_

#include "config.h"
#include "dataset.h"

#include <string>

namespace migbm {

class ObjectiveFunction {
 public:
  virtual ~ObjectiveFunction() = default;
  virtual void Init(const Metadata& metadata, data_size_t num_data) = 0;
  /*! grad/hess of loss wrt score, for all rows (score layout class-major for multiclass) */
  virtual void GetGradients(const double* score, score_t* gradients,
                            score_t* hessians) const = 0;
  virtual const char* GetName() const = 0;
  virtual std::string ToString() const { return GetName(); }
  virtual bool IsConstantHessian() const { return false; }
  virtual double BoostFromScore(int /*class_id*/) const { return 0.0; }
  virtual bool ClassNeedTrain(int /*class_id*/) const { return true; }
  virtual int NumModelPerIteration() const { return 1; }
  virtual int NumPredictOneRow() const { return 1; }
  virtual void ConvertOutput(const double* input, double* output) const { *output = *input; }
  virtual bool NeedConvertOutputCUDA() const { return false; }
  virtual bool NeedRenewTreeOutput() const { return false; }
  virtual double RenewTreeOutput(double orig_output, const data_size_t* indices,
                                 data_size_t cnt, const double* score) const {
    (void)indices; (void)cnt; (void)score;
    return orig_output;
  }
  virtual bool IsRenewTreeOutput() const { return NeedRenewTreeOutput(); }
  virtual data_size_t NumPositiveData() const { return 0; }
  /*! average output flag (RF / some objectives) */
  virtual bool average_output() const { return false; }

  static ObjectiveFunction* Create(const std::string& name, const Config& config);
  static ObjectiveFunction* CreateFromModelString(const std::string& str);
};

}  // namespace migbm

#endif  // MIGBM_OBJECTIVE_H_
