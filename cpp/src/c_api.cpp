/*! migbm C API implementation. Parity target: reference src/c_api.cpp — the Booster
 *  wrapper, dataset ingestion paths (mat/CSR/CSC/file) and prediction entry points. */
#include "migbm/c_api.h"

#include "migbm/arrow.h"

#include "migbm/boosting.h"
#include "migbm/common.h"
#include "migbm/config.h"
#include "migbm/dataset.h"
#include "migbm/metric.h"
#include "migbm/network.h"
#include "migbm/objective.h"

#include <mutex>
#include <string>

namespace migbm {

static thread_local std::string g_last_error = "Everything is fine";

#define API_BEGIN() try {
#define API_END()                         \
  return 0;                               \
  }                                       \
  catch (std::exception & ex) {           \
    migbm::g_last_error = ex.what();      \
    return -1;                            \
  }                                       \
  catch (...) {                           \
    migbm::g_last_error = "unknown exception"; \
    return -1;                            \
  }

/*! value accessor over the 4 supported dtypes */
static std::function<double(int64_t)> MakeGetter(const void* data, int data_type) {
  switch (data_type) {
    case C_API_DTYPE_FLOAT32: {
      const float* p = static_cast<const float*>(data);
      return [p](int64_t i) { return static_cast<double>(p[i]); };
    }
    case C_API_DTYPE_FLOAT64: {
      const double* p = static_cast<const double*>(data);
      return [p](int64_t i) { return p[i]; };
    }
    case C_API_DTYPE_INT32: {
      const int32_t* p = static_cast<const int32_t*>(data);
      return [p](int64_t i) { return static_cast<double>(p[i]); };
    }
    case C_API_DTYPE_INT64: {
      const int64_t* p = static_cast<const int64_t*>(data);
      return [p](int64_t i) { return static_cast<double>(p[i]); };
    }
  }
  Log::Fatal("Unknown data type %d", data_type);
  return nullptr;
}

static std::function<int64_t(int64_t)> MakeIndptrGetter(const void* p, int t) {
  if (t == C_API_DTYPE_INT32) {
    const int32_t* q = static_cast<const int32_t*>(p);
    return [q](int64_t i) { return static_cast<int64_t>(q[i]); };
  }
  const int64_t* q = static_cast<const int64_t*>(p);
  return [q](int64_t i) { return q[i]; };
}

static std::vector<int8_t> ParseCategoricalFlags(const Config& cfg, int ncol) {
  std::vector<int8_t> flags(ncol, 0);
  if (cfg.categorical_feature.empty()) return flags;
  for (auto& tok : Common::Split(cfg.categorical_feature.c_str(), ',')) {
    auto t = Common::Trim(tok);
    if (t.empty()) continue;
    int idx = atoi(t.c_str());
    if (idx >= 0 && idx < ncol) flags[idx] = 1;
  }
  return flags;
}

/*! Booster wrapper (parity: reference c_api.cpp Booster class). */
class BoosterWrapper {
 public:
  BoosterWrapper(const Dataset* train_data, const char* parameters) : train_data_(train_data) {
    auto params = Config::Str2Map(parameters);
    config_.Set(params);
    if (config_.num_threads > 0) omp_set_num_threads(config_.num_threads);
    objective_.reset(ObjectiveFunction::Create(config_.objective, config_));
    if (objective_ != nullptr && train_data_ != nullptr)
      objective_->Init(train_data_->metadata(), train_data_->num_data());
    CreateMetrics();
    boosting_.reset(GBDT::CreateBoosting(config_.boosting, nullptr));
    std::vector<const Metric*> train_m;
    for (auto& m : train_metrics_) train_m.push_back(m.get());
    boosting_->Init(&config_, train_data_, objective_.get(), train_m);
  }
  explicit BoosterWrapper(const char* model_filename) {
    boosting_.reset(GBDT::CreateBoosting("gbdt", model_filename));
  }
  BoosterWrapper(const char* model_str, size_t len) {
    boosting_.reset(GBDT::CreateBoosting("gbdt", nullptr));
    boosting_->LoadModelFromString(model_str, len);
  }

  void CreateMetrics() {
    train_metrics_.clear();
    auto metric_names = config_.metric;
    if (metric_names.empty() && !config_.objective.empty() && config_.objective != "none") {
      metric_names.push_back(config_.objective == "regression" ? "l2" : config_.objective);
    }
    for (auto& name : metric_names) {
      if (name == "none" || name == "null" || name == "na" || name == "custom") continue;
      std::unique_ptr<Metric> m(Metric::Create(name, config_));
      if (m != nullptr && train_data_ != nullptr) {
        m->Init(train_data_->metadata(), train_data_->num_data());
        train_metrics_.push_back(std::move(m));
      }
    }
  }

  void AddValidData(const Dataset* valid) {
    if (train_data_ != nullptr && !valid->AlignsWith(train_data_)) {
      Log::Fatal("Cannot add validation data: its bin mappers differ from the "
                 "training data's. Construct the valid set with the training "
                 "Dataset as reference (Dataset(..., reference=train_set)).");
    }
    std::vector<std::unique_ptr<Metric>> ms;
    auto metric_names = config_.metric;
    if (metric_names.empty() && !config_.objective.empty() && config_.objective != "none")
      metric_names.push_back(config_.objective == "regression" ? "l2" : config_.objective);
    for (auto& name : metric_names) {
      if (name == "none" || name == "null" || name == "na" || name == "custom") continue;
      std::unique_ptr<Metric> m(Metric::Create(name, config_));
      if (m != nullptr) {
        m->Init(valid->metadata(), valid->num_data());
        ms.push_back(std::move(m));
      }
    }
    std::vector<const Metric*> mp;
    for (auto& m : ms) mp.push_back(m.get());
    valid_metrics_.push_back(std::move(ms));
    boosting_->AddValidDataset(valid, mp);
  }

  void ResetParameter(const char* parameters) {
    auto params = Config::Str2Map(parameters);
    config_.Set(params);
    boosting_->ResetConfig(&config_);
  }

  void ResetTrainingData(const Dataset* train_data) {
    train_data_ = train_data;
    if (objective_ != nullptr)
      objective_->Init(train_data_->metadata(), train_data_->num_data());
    CreateMetrics();
    std::vector<const Metric*> train_m;
    for (auto& m : train_metrics_) train_m.push_back(m.get());
    boosting_->ResetTrainingData(train_data_, objective_.get(), train_m);
  }

  GBDT* boosting() { return boosting_.get(); }
  const Config& config() const { return config_; }
  std::mutex& mutex() { return mutex_; }

 private:
  const Dataset* train_data_ = nullptr;
  Config config_;
  std::unique_ptr<ObjectiveFunction> objective_;
  std::vector<std::unique_ptr<Metric>> train_metrics_;
  std::vector<std::vector<std::unique_ptr<Metric>>> valid_metrics_;
  std::unique_ptr<GBDT> boosting_;
  std::mutex mutex_;
};

static void PredictRows(GBDT* b, const std::function<void(int64_t, double*)>& row_getter,
                        int64_t nrow, int ncol, int predict_type, int start_iter, int num_iter,
                        double* out, const char* parameter = nullptr) {
  const int per_row = b->NumPredictOneRow(start_iter, num_iter,
                                          predict_type == C_API_PREDICT_LEAF_INDEX,
                                          predict_type == C_API_PREDICT_CONTRIB);
  // prediction early stopping (reference pred_early_stop params): skip remaining
  // iterations once the margin clears the threshold at a check round
  bool early_stop = false;
  int es_freq = 10;
  double es_margin = 10.0;
  if (parameter != nullptr && *parameter != '\0') {
    for (auto& kv : Config::Str2Map(parameter)) {
      if (kv.first == "pred_early_stop") early_stop = kv.second == "true" || kv.second == "1";
      else if (kv.first == "pred_early_stop_freq") es_freq = atoi(kv.second.c_str());
      else if (kv.first == "pred_early_stop_margin") es_margin = atof(kv.second.c_str());
    }
  }
  const bool es_usable = early_stop && (predict_type == C_API_PREDICT_NORMAL ||
                                        predict_type == C_API_PREDICT_RAW_SCORE);
#pragma omp parallel
  {
    std::vector<double> features(ncol);
#pragma omp for schedule(static)
    for (int64_t i = 0; i < nrow; ++i) {
      row_getter(i, features.data());
      double* o = out + i * per_row;
      if (es_usable) {
        b->PredictRawEarlyStop(features.data(), o, start_iter, num_iter, es_freq,
                               es_margin, per_row > 1);
        if (predict_type == C_API_PREDICT_NORMAL) b->ConvertRawToOutput(o);
        continue;
      }
      switch (predict_type) {
        case C_API_PREDICT_NORMAL: b->Predict(features.data(), o, start_iter, num_iter); break;
        case C_API_PREDICT_RAW_SCORE: b->PredictRaw(features.data(), o, start_iter, num_iter); break;
        case C_API_PREDICT_LEAF_INDEX: b->PredictLeafIndex(features.data(), o, start_iter, num_iter); break;
        case C_API_PREDICT_CONTRIB: b->PredictContrib(features.data(), o, start_iter, num_iter); break;
      }
    }
  }
}

static int CopyToBuffer(const std::string& s, int64_t buffer_len, int64_t* out_len,
                        char* out_str) {
  *out_len = static_cast<int64_t>(s.size()) + 1;
  if (buffer_len >= *out_len && out_str != nullptr) {
    memcpy(out_str, s.c_str(), *out_len);
  }
  return 0;
}

static int CopyStringsToBuffer(const std::vector<std::string>& strs, int len, int* out_len,
                               size_t buffer_len, size_t* out_buffer_len, char** out_strs) {
  *out_len = static_cast<int>(strs.size());
  size_t max_len = 1;
  for (auto& s : strs) max_len = std::max(max_len, s.size() + 1);
  *out_buffer_len = max_len;
  if (out_strs != nullptr && len >= static_cast<int>(strs.size())) {
    for (size_t i = 0; i < strs.size(); ++i) {
      if (buffer_len >= strs[i].size() + 1) {
        memcpy(out_strs[i], strs[i].c_str(), strs[i].size() + 1);
      }
    }
  }
  return 0;
}

}  // namespace migbm

using namespace migbm;

// ================================================================== misc
const char* LGBM_GetLastError() { return g_last_error.c_str(); }

int LGBM_RegisterLogCallback(void (*callback)(const char*)) {
  API_BEGIN();
  Log::Cb() = callback;
  API_END();
}

int LGBM_SetMaxThreads(int num_threads) {
  API_BEGIN();
  if (num_threads > 0) omp_set_num_threads(num_threads);
  API_END();
}

int LGBM_GetMaxThreads(int* out) {
  API_BEGIN();
  *out = omp_get_max_threads();
  API_END();
}

int LGBM_SetLastError(const char* msg) {
  migbm::g_last_error = msg != nullptr ? msg : "";
  return 0;
}

int LGBM_DumpParamAliases(int64_t buffer_len, int64_t* out_len, char* out_str) {
  API_BEGIN();
  std::stringstream ss;
  ss << "{";
  bool first = true;
  std::map<std::string, std::vector<std::string>> by_canonical;
  for (auto& kv : Config::alias_table()) by_canonical[kv.second].push_back(kv.first);
  for (auto& kv : by_canonical) {
    if (!first) ss << ",";
    first = false;
    ss << "\"" << kv.first << "\":[";
    for (size_t i = 0; i < kv.second.size(); ++i) {
      if (i) ss << ",";
      ss << "\"" << kv.second[i] << "\"";
    }
    ss << "]";
  }
  ss << "}";
  CopyToBuffer(ss.str(), buffer_len, out_len, out_str);
  API_END();
}

int LGBM_GetSampleCount(int32_t num_total_row, const char* parameters, int* out) {
  API_BEGIN();
  Config cfg;
  cfg.Set(Config::Str2Map(parameters));
  *out = std::min(num_total_row, cfg.bin_construct_sample_cnt);
  API_END();
}

int LGBM_SampleIndices(int32_t num_total_row, const char* parameters, void* out,
                       int32_t* out_len) {
  API_BEGIN();
  Config cfg;
  cfg.Set(Config::Str2Map(parameters));
  int cnt = std::min(num_total_row, cfg.bin_construct_sample_cnt);
  Random rng(cfg.data_random_seed);
  auto idx = rng.Sample(num_total_row, cnt);
  int32_t* o = static_cast<int32_t*>(out);
  for (size_t i = 0; i < idx.size(); ++i) o[i] = idx[i];
  *out_len = static_cast<int32_t>(idx.size());
  API_END();
}

// ================================================================== dataset
int LGBM_DatasetCreateFromMat(const void* data, int data_type, int32_t nrow, int32_t ncol,
                              int is_row_major, const char* parameters,
                              const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  auto get = MakeGetter(data, data_type);
  std::function<double(data_size_t, int)> at;
  if (is_row_major) {
    at = [get, ncol](data_size_t r, int c) { return get(static_cast<int64_t>(r) * ncol + c); };
  } else {
    at = [get, nrow](data_size_t r, int c) { return get(static_cast<int64_t>(c) * nrow + r); };
  }
  if (reference != nullptr) {
    const Dataset* ref = static_cast<const Dataset*>(reference);
    *out = ref->CreateValid(at, nrow).release();
  } else {
    Config cfg;
    cfg.Set(Config::Str2Map(parameters));
    auto d = std::make_unique<Dataset>();
    d->ConstructFromMat(at, nrow, ncol, cfg, ParseCategoricalFlags(cfg, ncol));
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetCreateFromMats(int32_t nmat, const void** data, int data_type, int32_t* nrows,
                               int32_t ncol, int is_row_major, const char* parameters,
                               const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  MIGBM_CHECK(is_row_major);
  int64_t total = 0;
  std::vector<int64_t> starts(nmat);
  std::vector<std::function<double(int64_t)>> getters(nmat);
  for (int m = 0; m < nmat; ++m) {
    starts[m] = total;
    total += nrows[m];
    getters[m] = MakeGetter(data[m], data_type);
  }
  auto at = [&](data_size_t r, int c) -> double {
    int m = 0;
    while (m + 1 < nmat && r >= starts[m + 1]) ++m;
    return getters[m]((static_cast<int64_t>(r) - starts[m]) * ncol + c);
  };
  if (reference != nullptr) {
    const Dataset* ref = static_cast<const Dataset*>(reference);
    *out = ref->CreateValid(at, static_cast<data_size_t>(total)).release();
  } else {
    Config cfg;
    cfg.Set(Config::Str2Map(parameters));
    auto d = std::make_unique<Dataset>();
    d->ConstructFromMat(at, static_cast<data_size_t>(total), ncol, cfg,
                        ParseCategoricalFlags(cfg, ncol));
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetCreateFromCSR(const void* indptr, int indptr_type, const int32_t* indices,
                              const void* data, int data_type, int64_t nindptr, int64_t nelem,
                              int64_t num_col, const char* parameters,
                              const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  (void)nelem;
  auto ip = MakeIndptrGetter(indptr, indptr_type);
  auto val = MakeGetter(data, data_type);
  const data_size_t nrow = static_cast<data_size_t>(nindptr - 1);
  // densify per-row on access (binning samples columns; fine for moderate ncol)
  auto at = [&, ip, val, indices](data_size_t r, int c) -> double {
    int64_t s = ip(r), e = ip(r + 1);
    // binary search on indices
    int64_t lo = s, hi = e - 1;
    while (lo <= hi) {
      int64_t mid = (lo + hi) >> 1;
      if (indices[mid] == c) return val(mid);
      if (indices[mid] < c) lo = mid + 1;
      else hi = mid - 1;
    }
    return 0.0;
  };
  if (reference != nullptr) {
    const Dataset* ref = static_cast<const Dataset*>(reference);
    *out = ref->CreateValid(at, nrow).release();
  } else {
    Config cfg;
    cfg.Set(Config::Str2Map(parameters));
    auto d = std::make_unique<Dataset>();
    d->ConstructFromMat(at, nrow, static_cast<int>(num_col), cfg,
                        ParseCategoricalFlags(cfg, static_cast<int>(num_col)));
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetCreateFromCSC(const void* col_ptr, int col_ptr_type, const int32_t* indices,
                              const void* data, int data_type, int64_t ncol_ptr, int64_t nelem,
                              int64_t num_row, const char* parameters,
                              const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  (void)nelem;
  auto cp = MakeIndptrGetter(col_ptr, col_ptr_type);
  auto val = MakeGetter(data, data_type);
  const int ncol = static_cast<int>(ncol_ptr - 1);
  auto at = [&, cp, val, indices](data_size_t r, int c) -> double {
    int64_t s = cp(c), e = cp(c + 1);
    int64_t lo = s, hi = e - 1;
    while (lo <= hi) {
      int64_t mid = (lo + hi) >> 1;
      if (indices[mid] == static_cast<int32_t>(r)) return val(mid);
      if (indices[mid] < static_cast<int32_t>(r)) lo = mid + 1;
      else hi = mid - 1;
    }
    return 0.0;
  };
  if (reference != nullptr) {
    const Dataset* ref = static_cast<const Dataset*>(reference);
    *out = ref->CreateValid(at, static_cast<data_size_t>(num_row)).release();
  } else {
    Config cfg;
    cfg.Set(Config::Str2Map(parameters));
    auto d = std::make_unique<Dataset>();
    d->ConstructFromMat(at, static_cast<data_size_t>(num_row), ncol, cfg,
                        ParseCategoricalFlags(cfg, ncol));
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetCreateFromFile(const char* filename, const char* parameters,
                               const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  Config cfg;
  cfg.Set(Config::Str2Map(parameters));
  if (Dataset::IsBinFile(filename)) {
    *out = Dataset::LoadFromBinFile(filename).release();
  } else {
    DatasetLoader loader(cfg);
    if (reference != nullptr) {
      *out = loader.LoadFromFileAlignWithOtherDataset(
                     filename, static_cast<const Dataset*>(reference)).release();
    } else {
      *out = loader.LoadFromFile(filename, 0, cfg.pre_partition ? cfg.num_machines : 1).release();
    }
  }
  API_END();
}

int LGBM_DatasetCreateFromSampledColumn(double** sample_data, int** sample_indices,
                                        int32_t ncol, const int* num_per_col,
                                        int32_t num_sample_row, int32_t num_local_row,
                                        int64_t /*num_dist_row*/, const char* parameters,
                                        DatasetHandle* out) {
  API_BEGIN();
  Config cfg;
  cfg.Set(Config::Str2Map(parameters));
  auto d = std::make_unique<Dataset>();
  d->ConstructFromSampleData(sample_data, sample_indices, ncol, num_per_col,
                             num_sample_row, num_local_row, cfg,
                             ParseCategoricalFlags(cfg, ncol));
  *out = d.release();
  API_END();
}

int LGBM_DatasetCreateByReference(const DatasetHandle reference, int64_t num_total_row,
                                  DatasetHandle* out) {
  API_BEGIN();
  const Dataset* ref = static_cast<const Dataset*>(reference);
  *out = ref->CreateByReference(static_cast<data_size_t>(num_total_row)).release();
  API_END();
}

int LGBM_DatasetInitStreaming(DatasetHandle, int32_t, int32_t, int32_t, int32_t, int32_t,
                              int32_t) {
  API_BEGIN();
  // streaming state is implicit in this build (columns are pre-sized; pushes are
  // positional and thread-safe per disjoint row ranges)
  API_END();
}

int LGBM_DatasetPushRows(DatasetHandle dataset, const void* data, int data_type,
                         int32_t nrow, int32_t ncol, int32_t start_row) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(dataset);
  auto get = MakeGetter(data, data_type);
  std::vector<double> row(ncol);
#pragma omp parallel for schedule(static) private(row)
  for (int32_t i = 0; i < nrow; ++i) {
    std::vector<double> r(ncol);
    for (int c = 0; c < ncol; ++c) r[c] = get(static_cast<int64_t>(i) * ncol + c);
    d->PushRawRow(start_row + i, r.data(), ncol);
  }
  API_END();
}

int LGBM_DatasetPushRowsWithMetadata(DatasetHandle dataset, const void* data, int data_type,
                                     int32_t nrow, int32_t ncol, int32_t start_row,
                                     const float* labels, const float* weights,
                                     const double* init_scores, const int32_t* queries,
                                     int32_t /*tid*/) {
  API_BEGIN();
  int rc = LGBM_DatasetPushRows(dataset, data, data_type, nrow, ncol, start_row);
  if (rc != 0) return rc;
  Dataset* d = static_cast<Dataset*>(dataset);
  if (labels != nullptr) {
    auto& lab = d->metadata().mutable_label();
    for (int32_t i = 0; i < nrow; ++i) lab[start_row + i] = labels[i];
  }
  if (weights != nullptr) d->metadata().SetStreamedWeights(start_row, weights, nrow);
  if (init_scores != nullptr) d->metadata().SetStreamedInitScores(start_row, init_scores, nrow);
  if (queries != nullptr) d->metadata().SetStreamedQueryIds(start_row, queries, nrow);
  API_END();
}

int LGBM_DatasetCreateFromCSRFunc(void* get_row_funptr, int num_rows, int64_t num_col,
                                  const char* parameters, const DatasetHandle reference,
                                  DatasetHandle* out) {
  API_BEGIN();
  // row provider is a std::function supplying (index, value) pairs per row
  // (reference c_api.cpp semantics for the SWIG/R ingestion path)
  using RowFunc = std::function<void(int idx, std::vector<std::pair<int, double>>&)>;
  RowFunc* get_row = reinterpret_cast<RowFunc*>(get_row_funptr);
  std::vector<std::vector<std::pair<int, double>>> rows(num_rows);
  for (int i = 0; i < num_rows; ++i) (*get_row)(i, rows[i]);
  auto getter = [&rows](data_size_t r, int c) -> double {
    for (auto& kv : rows[r])
      if (kv.first == c) return kv.second;
    return 0.0;
  };
  Config cfg;
  cfg.Set(Config::Str2Map(parameters));
  if (reference != nullptr) {
    *out = static_cast<const Dataset*>(reference)
               ->CreateValid(getter, num_rows).release();
  } else {
    auto d = std::make_unique<Dataset>(num_rows);
    d->ConstructFromMat(getter, num_rows, static_cast<int>(num_col), cfg,
                        ParseCategoricalFlags(cfg, static_cast<int>(num_col)));
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetPushRowsByCSRWithMetadata(DatasetHandle dataset, const void* indptr,
                                          int indptr_type, const int32_t* indices,
                                          const void* data, int data_type, int64_t nindptr,
                                          int64_t nelem, int64_t start_row,
                                          const float* labels, const float* weights,
                                          const double* init_scores,
                                          const int32_t* queries, int32_t /*tid*/) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(dataset);
  const int64_t num_col = d->num_total_features();
  int rc = LGBM_DatasetPushRowsByCSR(dataset, indptr, indptr_type, indices, data, data_type,
                                     nindptr, nelem, num_col, start_row);
  if (rc != 0) return rc;
  const int32_t nrow = static_cast<int32_t>(nindptr - 1);
  if (labels != nullptr) {
    auto& lab = d->metadata().mutable_label();
    for (int32_t i = 0; i < nrow; ++i) lab[start_row + i] = labels[i];
  }
  if (weights != nullptr) d->metadata().SetStreamedWeights(start_row, weights, nrow);
  if (init_scores != nullptr) d->metadata().SetStreamedInitScores(start_row, init_scores, nrow);
  if (queries != nullptr) d->metadata().SetStreamedQueryIds(start_row, queries, nrow);
  API_END();
}

int LGBM_DatasetPushRowsByCSR(DatasetHandle dataset, const void* indptr, int indptr_type,
                              const int32_t* indices, const void* data, int data_type,
                              int64_t nindptr, int64_t /*nelem*/, int64_t num_col,
                              int64_t start_row) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(dataset);
  auto ip = MakeIndptrGetter(indptr, indptr_type);
  auto val = MakeGetter(data, data_type);
  const int32_t nrow = static_cast<int32_t>(nindptr - 1);
#pragma omp parallel for schedule(static)
  for (int32_t i = 0; i < nrow; ++i) {
    std::vector<double> r(num_col, 0.0);
    for (int64_t k = ip(i); k < ip(i + 1); ++k)
      if (indices[k] < num_col) r[indices[k]] = val(k);
    d->PushRawRow(static_cast<data_size_t>(start_row + i), r.data(),
                  static_cast<int>(num_col));
  }
  API_END();
}

int LGBM_DatasetMarkFinished(DatasetHandle dataset) {
  API_BEGIN();
  static_cast<Dataset*>(dataset)->metadata().FinalizeStreamedQueries();
  API_END();
}

int LGBM_DatasetSetWaitForManualFinish(DatasetHandle, int) {
  API_BEGIN();
  API_END();
}

int LGBM_DatasetSerializeReferenceToBinary(DatasetHandle handle, ByteBufferHandle* out,
                                           int32_t* out_len) {
  API_BEGIN();
  auto* buf = new std::vector<uint8_t>();
  auto s = static_cast<Dataset*>(handle)->SerializeReference();
  buf->assign(s.begin(), s.end());
  *out = buf;
  *out_len = static_cast<int32_t>(buf->size());
  API_END();
}

int LGBM_ByteBufferGetAt(ByteBufferHandle handle, int32_t index, uint8_t* out_val) {
  API_BEGIN();
  auto* buf = static_cast<std::vector<uint8_t>*>(handle);
  if (index < 0 || index >= static_cast<int32_t>(buf->size()))
    Log::Fatal("ByteBuffer index %d out of range", index);
  *out_val = (*buf)[index];
  API_END();
}

int LGBM_ByteBufferFree(ByteBufferHandle handle) {
  API_BEGIN();
  delete static_cast<std::vector<uint8_t>*>(handle);
  API_END();
}

int LGBM_DatasetCreateFromSerializedReference(const void* ref_buffer,
                                              int32_t ref_buffer_size, int64_t num_row,
                                              int32_t /*num_classes*/,
                                              const char* /*parameters*/,
                                              DatasetHandle* out) {
  API_BEGIN();
  *out = Dataset::FromSerializedReference(static_cast<const char*>(ref_buffer),
                                          ref_buffer_size,
                                          static_cast<data_size_t>(num_row)).release();
  API_END();
}

// ------------------------------------------------------------------ Arrow ingestion
namespace {

std::function<double(int64_t)> ArrowColGetter(const ArrowArray* a, const char* fmt) {
  const uint8_t* validity = a->n_buffers > 0 ? static_cast<const uint8_t*>(a->buffers[0])
                                             : nullptr;
  const void* data = a->n_buffers > 1 ? a->buffers[1] : nullptr;
  const int64_t off = a->offset;
  auto is_null = [validity, off](int64_t i) {
    if (validity == nullptr) return false;
    const int64_t j = i + off;
    return ((validity[j >> 3] >> (j & 7)) & 1) == 0;
  };
  const double kNaN = std::numeric_limits<double>::quiet_NaN();
  switch (fmt[0]) {
    case 'g': {
      const double* p = static_cast<const double*>(data);
      return [p, off, is_null, kNaN](int64_t i) { return is_null(i) ? kNaN : p[i + off]; };
    }
    case 'f': {
      const float* p = static_cast<const float*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'l': {
      const int64_t* p = static_cast<const int64_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'L': {
      const uint64_t* p = static_cast<const uint64_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'i': {
      const int32_t* p = static_cast<const int32_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'I': {
      const uint32_t* p = static_cast<const uint32_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 's': {
      const int16_t* p = static_cast<const int16_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'c': {
      const int8_t* p = static_cast<const int8_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'C': {
      const uint8_t* p = static_cast<const uint8_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        return is_null(i) ? kNaN : static_cast<double>(p[i + off]);
      };
    }
    case 'b': {
      const uint8_t* p = static_cast<const uint8_t*>(data);
      return [p, off, is_null, kNaN](int64_t i) {
        if (is_null(i)) return kNaN;
        const int64_t j = i + off;
        return static_cast<double>((p[j >> 3] >> (j & 7)) & 1);
      };
    }
  }
  Log::Fatal("Unsupported Arrow column format '%s'", fmt);
  return nullptr;
}

}  // namespace

int LGBM_DatasetCreateFromArrow(int64_t n_chunks, const struct ArrowArray* chunks,
                                const struct ArrowSchema* schema, const char* parameters,
                                const DatasetHandle reference, DatasetHandle* out) {
  API_BEGIN();
  MIGBM_CHECK(schema->n_children > 0);
  const int ncol = static_cast<int>(schema->n_children);
  // per-chunk per-column getters + chunk row offsets
  std::vector<int64_t> chunk_start(n_chunks + 1, 0);
  std::vector<std::vector<std::function<double(int64_t)>>> getters(n_chunks);
  for (int64_t c = 0; c < n_chunks; ++c) {
    chunk_start[c + 1] = chunk_start[c] + chunks[c].length;
    MIGBM_CHECK_EQ(chunks[c].n_children, schema->n_children);
    getters[c].resize(ncol);
    for (int f = 0; f < ncol; ++f)
      getters[c][f] = ArrowColGetter(chunks[c].children[f], schema->children[f]->format);
  }
  const data_size_t nrow = static_cast<data_size_t>(chunk_start[n_chunks]);
  auto at = [&](data_size_t r, int col) -> double {
    // chunk lookup (chunks are few; linear scan with memo would do, binary search is fine)
    int64_t lo = 0, hi = n_chunks - 1;
    while (lo < hi) {
      const int64_t mid = (lo + hi + 1) >> 1;
      if (r >= chunk_start[mid]) lo = mid;
      else hi = mid - 1;
    }
    return getters[lo][col](r - chunk_start[lo]);
  };
  if (reference != nullptr) {
    *out = static_cast<const Dataset*>(reference)->CreateValid(at, nrow).release();
  } else {
    Config cfg;
    cfg.Set(Config::Str2Map(parameters));
    auto d = std::make_unique<Dataset>();
    // column names from the schema
    std::vector<std::string> names;
    for (int f = 0; f < ncol; ++f)
      names.push_back(schema->children[f]->name ? schema->children[f]->name
                                                : "Column_" + std::to_string(f));
    d->ConstructFromMat(at, nrow, ncol, cfg, ParseCategoricalFlags(cfg, ncol));
    d->set_feature_names(names);
    *out = d.release();
  }
  API_END();
}

int LGBM_DatasetSetFieldFromArrow(DatasetHandle handle, const char* field_name,
                                  int64_t n_chunks, const struct ArrowArray* chunks,
                                  const struct ArrowSchema* schema) {
  API_BEGIN();
  // single flat numeric column expected
  int64_t total = 0;
  for (int64_t c = 0; c < n_chunks; ++c) total += chunks[c].length;
  std::vector<double> vals;
  vals.reserve(total);
  for (int64_t c = 0; c < n_chunks; ++c) {
    auto get = ArrowColGetter(&chunks[c], schema->format);
    for (int64_t i = 0; i < chunks[c].length; ++i) vals.push_back(get(i));
  }
  std::string name(field_name);
  Dataset* d = static_cast<Dataset*>(handle);
  if (name == "label" || name == "weight") {
    std::vector<float> f(vals.begin(), vals.end());
    if (name == "label") d->metadata().SetLabel(f.data(), static_cast<data_size_t>(f.size()));
    else d->metadata().SetWeights(f.data(), static_cast<data_size_t>(f.size()));
  } else if (name == "group" || name == "query") {
    std::vector<int32_t> g(vals.begin(), vals.end());
    d->metadata().SetQuery(g.data(), static_cast<data_size_t>(g.size()));
  } else if (name == "init_score") {
    d->metadata().SetInitScore(vals.data(), static_cast<int64_t>(vals.size()));
  } else {
    Log::Fatal("Unknown field %s", field_name);
  }
  API_END();
}

int LGBM_DatasetGetSubset(const DatasetHandle handle, const int32_t* used_row_indices,
                          int32_t num_used_row_indices, const char* /*parameters*/,
                          DatasetHandle* out) {
  API_BEGIN();
  const Dataset* d = static_cast<const Dataset*>(handle);
  *out = d->Subset(used_row_indices, num_used_row_indices).release();
  API_END();
}

int LGBM_DatasetSetFeatureNames(DatasetHandle handle, const char** feature_names, int num) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(handle);
  std::vector<std::string> names;
  for (int i = 0; i < num; ++i) names.emplace_back(feature_names[i]);
  d->set_feature_names(names);
  API_END();
}

int LGBM_DatasetGetFeatureNames(DatasetHandle handle, const int len, int* num_feature_names,
                                const size_t buffer_len, size_t* out_buffer_len,
                                char** feature_names) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(handle);
  CopyStringsToBuffer(d->feature_names(), len, num_feature_names, buffer_len, out_buffer_len,
                      feature_names);
  API_END();
}

int LGBM_DatasetFree(DatasetHandle handle) {
  API_BEGIN();
  delete static_cast<Dataset*>(handle);
  API_END();
}

int LGBM_DatasetSaveBinary(DatasetHandle handle, const char* filename) {
  API_BEGIN();
  static_cast<Dataset*>(handle)->SaveBinaryFile(filename);
  API_END();
}

int LGBM_DatasetDumpText(DatasetHandle handle, const char* filename) {
  API_BEGIN();
  static_cast<Dataset*>(handle)->DumpTextFile(filename);
  API_END();
}

int LGBM_DatasetAddFeaturesFrom(DatasetHandle target, DatasetHandle source) {
  API_BEGIN();
  static_cast<Dataset*>(target)->AddFeaturesFrom(static_cast<Dataset*>(source));
  API_END();
}

int LGBM_DatasetSetField(DatasetHandle handle, const char* field_name, const void* field_data,
                         int num_element, int type) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(handle);
  std::string name(field_name);
  // a null/empty payload CLEARS an optional field (reference set_field(None))
  const bool clearing = field_data == nullptr || num_element == 0;
  if (clearing && name != "label") {
    if (name == "weight") d->metadata().SetWeights(nullptr, 0);
    else if (name == "group" || name == "query") d->metadata().SetQuery(nullptr, 0);
    else if (name == "init_score") d->metadata().SetInitScore(nullptr, 0);
    else if (name == "position") d->metadata().SetPosition(nullptr, 0);
    else Log::Fatal("Unknown field %s", field_name);
    return 0;
  }
  if ((name == "label" || name == "weight" || name == "position") &&
      num_element != d->num_data()) {
    Log::Fatal("Length of %s (%d) differs from the number of rows (%d)", field_name,
               num_element, d->num_data());
  }
  if (name == "label" || name == "weight") {
    std::vector<float> buf(num_element);
    if (type == C_API_DTYPE_FLOAT32) {
      const float* p = static_cast<const float*>(field_data);
      std::copy(p, p + num_element, buf.begin());
    } else if (type == C_API_DTYPE_FLOAT64) {
      const double* p = static_cast<const double*>(field_data);
      for (int i = 0; i < num_element; ++i) buf[i] = static_cast<float>(p[i]);
    } else {
      Log::Fatal("Bad type for %s", field_name);
    }
    if (name == "label") d->metadata().SetLabel(buf.data(), num_element);
    else d->metadata().SetWeights(buf.data(), num_element);
  } else if (name == "group" || name == "query") {
    MIGBM_CHECK(type == C_API_DTYPE_INT32);
    d->metadata().SetQuery(static_cast<const int32_t*>(field_data), num_element);
  } else if (name == "init_score") {
    MIGBM_CHECK(type == C_API_DTYPE_FLOAT64);
    d->metadata().SetInitScore(static_cast<const double*>(field_data), num_element);
  } else if (name == "position") {
    MIGBM_CHECK(type == C_API_DTYPE_INT32);
    d->metadata().SetPosition(static_cast<const int32_t*>(field_data), num_element);
  } else {
    Log::Fatal("Unknown field %s", field_name);
  }
  API_END();
}

int LGBM_DatasetGetField(DatasetHandle handle, const char* field_name, int* out_len,
                         const void** out_ptr, int* out_type) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(handle);
  std::string name(field_name);
  if (name == "label") {
    *out_ptr = d->metadata().label();
    *out_len = d->num_data();
    *out_type = C_API_DTYPE_FLOAT32;
  } else if (name == "weight") {
    *out_ptr = d->metadata().weights();
    *out_len = *out_ptr ? d->num_data() : 0;
    *out_type = C_API_DTYPE_FLOAT32;
  } else if (name == "group") {
    *out_ptr = d->metadata().query_boundaries();
    *out_len = *out_ptr ? d->metadata().num_queries() + 1 : 0;
    *out_type = C_API_DTYPE_INT32;
  } else if (name == "init_score") {
    *out_ptr = d->metadata().init_score();
    *out_len = static_cast<int>(d->metadata().num_init_score());
    *out_type = C_API_DTYPE_FLOAT64;
  } else if (name == "position") {
    *out_ptr = d->metadata().positions();
    *out_len = *out_ptr ? d->num_data() : 0;
    *out_type = C_API_DTYPE_INT32;
  } else {
    Log::Fatal("Unknown field %s", field_name);
  }
  API_END();
}

int LGBM_DatasetUpdateParamChecking(const char*, const char*) {
  API_BEGIN();
  API_END();
}

int LGBM_DatasetGetNumData(DatasetHandle handle, int32_t* out) {
  API_BEGIN();
  *out = static_cast<Dataset*>(handle)->num_data();
  API_END();
}

int LGBM_DatasetGetNumFeature(DatasetHandle handle, int32_t* out) {
  API_BEGIN();
  *out = static_cast<Dataset*>(handle)->num_total_features();
  API_END();
}

int LGBM_DatasetGetFeatureNumBin(DatasetHandle handle, int feature, int32_t* out) {
  API_BEGIN();
  Dataset* d = static_cast<Dataset*>(handle);
  int inner = d->InnerFeatureIndex(feature);
  *out = inner >= 0 ? d->FeatureNumBin(inner) : 0;
  API_END();
}

// ================================================================== booster
int LGBM_BoosterCreate(const DatasetHandle train_data, const char* parameters,
                       BoosterHandle* out) {
  API_BEGIN();
  *out = new BoosterWrapper(static_cast<const Dataset*>(train_data), parameters);
  API_END();
}

int LGBM_BoosterCreateFromModelfile(const char* filename, int* out_num_iterations,
                                    BoosterHandle* out) {
  API_BEGIN();
  auto* b = new BoosterWrapper(filename);
  *out_num_iterations = b->boosting()->GetCurrentIteration();
  *out = b;
  API_END();
}

int LGBM_BoosterLoadModelFromString(const char* model_str, int* out_num_iterations,
                                    BoosterHandle* out) {
  API_BEGIN();
  auto* b = new BoosterWrapper(model_str, strlen(model_str));
  *out_num_iterations = b->boosting()->GetCurrentIteration();
  *out = b;
  API_END();
}

int LGBM_BoosterFree(BoosterHandle handle) {
  API_BEGIN();
  delete static_cast<BoosterWrapper*>(handle);
  API_END();
}

int LGBM_BoosterShuffleModels(BoosterHandle handle, int start_iter, int end_iter) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->boosting()->ShuffleModels(start_iter, end_iter);
  API_END();
}

int LGBM_BoosterMerge(BoosterHandle handle, BoosterHandle other_handle) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle);
  auto* o = static_cast<BoosterWrapper*>(other_handle);
  b->boosting()->MergeFrom(o->boosting());
  API_END();
}

int LGBM_BoosterAddValidData(BoosterHandle handle, const DatasetHandle valid_data) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->AddValidData(static_cast<const Dataset*>(valid_data));
  API_END();
}

int LGBM_BoosterResetTrainingData(BoosterHandle handle, const DatasetHandle train_data) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->ResetTrainingData(
      static_cast<const Dataset*>(train_data));
  API_END();
}

int LGBM_BoosterResetParameter(BoosterHandle handle, const char* parameters) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->ResetParameter(parameters);
  API_END();
}

int LGBM_BoosterGetNumClasses(BoosterHandle handle, int* out_len) {
  API_BEGIN();
  *out_len = static_cast<BoosterWrapper*>(handle)->boosting()->num_class();
  API_END();
}

int LGBM_BoosterUpdateOneIter(BoosterHandle handle, int* is_finished) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle);
  std::lock_guard<std::mutex> lock(b->mutex());
  *is_finished = b->boosting()->TrainOneIter(nullptr, nullptr) ? 1 : 0;
  API_END();
}

int LGBM_BoosterUpdateOneIterCustom(BoosterHandle handle, const float* grad, const float* hess,
                                    int* is_finished) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle);
  std::lock_guard<std::mutex> lock(b->mutex());
  *is_finished = b->boosting()->TrainOneIter(grad, hess) ? 1 : 0;
  API_END();
}

int LGBM_BoosterRefit(BoosterHandle handle, const int32_t* leaf_preds, int32_t nrow,
                      int32_t ncol) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle);
  std::lock_guard<std::mutex> lock(b->mutex());
  b->boosting()->RefitTree(leaf_preds, nrow, ncol);
  API_END();
}

int LGBM_BoosterRollbackOneIter(BoosterHandle handle) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle);
  std::lock_guard<std::mutex> lock(b->mutex());
  b->boosting()->RollbackOneIter();
  API_END();
}

int LGBM_BoosterGetCurrentIteration(BoosterHandle handle, int* out_iteration) {
  API_BEGIN();
  *out_iteration = static_cast<BoosterWrapper*>(handle)->boosting()->GetCurrentIteration();
  API_END();
}

int LGBM_BoosterNumModelPerIteration(BoosterHandle handle, int* out) {
  API_BEGIN();
  *out = static_cast<BoosterWrapper*>(handle)->boosting()->num_tree_per_iteration();
  API_END();
}

int LGBM_BoosterNumberOfTotalModel(BoosterHandle handle, int* out) {
  API_BEGIN();
  *out = static_cast<BoosterWrapper*>(handle)->boosting()->NumberOfTotalModel();
  API_END();
}

int LGBM_BoosterGetEvalCounts(BoosterHandle handle, int* out_len) {
  API_BEGIN();
  *out_len = static_cast<int>(
      static_cast<BoosterWrapper*>(handle)->boosting()->EvalNames().size());
  API_END();
}

int LGBM_BoosterGetEvalNames(BoosterHandle handle, const int len, int* out_len,
                             const size_t buffer_len, size_t* out_buffer_len,
                             char** out_strs) {
  API_BEGIN();
  CopyStringsToBuffer(static_cast<BoosterWrapper*>(handle)->boosting()->EvalNames(), len,
                      out_len, buffer_len, out_buffer_len, out_strs);
  API_END();
}

int LGBM_BoosterGetFeatureNames(BoosterHandle handle, const int len, int* out_len,
                                const size_t buffer_len, size_t* out_buffer_len,
                                char** out_strs) {
  API_BEGIN();
  CopyStringsToBuffer(static_cast<BoosterWrapper*>(handle)->boosting()->FeatureNames(), len,
                      out_len, buffer_len, out_buffer_len, out_strs);
  API_END();
}

int LGBM_BoosterValidateFeatureNames(BoosterHandle handle, const char** data_names,
                                     int data_num_features) {
  API_BEGIN();
  const auto& names = static_cast<BoosterWrapper*>(handle)->boosting()->FeatureNames();
  if (static_cast<int>(names.size()) != data_num_features)
    Log::Fatal("Expected %d features but data has %d features",
               static_cast<int>(names.size()), data_num_features);
  for (int i = 0; i < data_num_features; ++i) {
    if (names[i] != data_names[i])
      Log::Fatal("Expected '%s' at position %d but found '%s'", names[i].c_str(), i,
                 data_names[i]);
  }
  API_END();
}

int LGBM_BoosterGetLoadedParam(BoosterHandle handle, int64_t buffer_len, int64_t* out_len,
                               char* out_str) {
  API_BEGIN();
  // "[key: value]" lines of the model's parameters block rendered as JSON;
  // fresh (in-memory trained) boosters fall back to their live config echo
  auto* w = static_cast<BoosterWrapper*>(handle);
  std::string raw = w->boosting()->LoadedParameter();
  if (raw.empty()) raw = w->config().SaveHyperParameters();
  std::stringstream js;
  js << "{";
  bool first = true;
  for (auto& line : Common::Split(raw.c_str(), '\n')) {
    auto t = Common::Trim(line);
    if (t.size() < 3 || t.front() != '[' || t.back() != ']') continue;
    t = t.substr(1, t.size() - 2);
    auto colon = t.find(':');
    if (colon == std::string::npos) continue;
    std::string k = Common::Trim(t.substr(0, colon));
    std::string v = Common::Trim(t.substr(colon + 1));
    if (!first) js << ",";
    first = false;
    char* endp = nullptr;
    strtod(v.c_str(), &endp);
    const bool numeric = !v.empty() && endp != nullptr && *endp == '\0';
    js << "\"" << k << "\":";
    if (numeric) js << v;
    else js << "\"" << v << "\"";
  }
  js << "}";
  return CopyToBuffer(js.str(), buffer_len, out_len, out_str);
  API_END();
}

int LGBM_BoosterGetNumFeature(BoosterHandle handle, int* out_len) {
  API_BEGIN();
  *out_len = static_cast<BoosterWrapper*>(handle)->boosting()->MaxFeatureIdx() + 1;
  API_END();
}

int LGBM_BoosterGetEval(BoosterHandle handle, int data_idx, int* out_len,
                        double* out_results) {
  API_BEGIN();
  auto r = static_cast<BoosterWrapper*>(handle)->boosting()->GetEvalAt(data_idx);
  *out_len = static_cast<int>(r.size());
  std::copy(r.begin(), r.end(), out_results);
  API_END();
}

int LGBM_BoosterGetNumPredict(BoosterHandle handle, int data_idx, int64_t* out_len) {
  API_BEGIN();
  *out_len = static_cast<BoosterWrapper*>(handle)->boosting()->GetNumPredictAt(data_idx);
  API_END();
}

int LGBM_BoosterGetPredict(BoosterHandle handle, int data_idx, int64_t* out_len,
                           double* out_result) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->boosting()->GetPredictAt(data_idx, out_result, out_len);
  API_END();
}

int LGBM_BoosterCalcNumPredict(BoosterHandle handle, int num_row, int predict_type,
                               int start_iteration, int num_iteration, int64_t* out_len) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  *out_len = static_cast<int64_t>(num_row) *
             b->NumPredictOneRow(start_iteration, num_iteration,
                                 predict_type == C_API_PREDICT_LEAF_INDEX,
                                 predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

int LGBM_BoosterPredictForMat(BoosterHandle handle, const void* data, int data_type,
                              int32_t nrow, int32_t ncol, int is_row_major, int predict_type,
                              int start_iteration, int num_iteration, const char* parameter,
                              int64_t* out_len, double* out_result) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  bool disable_shape_check = false;
  if (parameter != nullptr) {
    for (auto& kv : Config::Str2Map(parameter))
      if (kv.first == "predict_disable_shape_check")
        disable_shape_check = kv.second == "true" || kv.second == "1";
  }
  if (!disable_shape_check && ncol <= b->MaxFeatureIdx())
    Log::Fatal("The number of features in data (%d) is fewer than it was in training data "
               "(%d)", ncol, b->MaxFeatureIdx() + 1);
  auto get = MakeGetter(data, data_type);
  std::function<void(int64_t, double*)> row_getter;
  if (is_row_major) {
    row_getter = [get, ncol](int64_t r, double* o) {
      for (int c = 0; c < ncol; ++c) o[c] = get(r * ncol + c);
    };
  } else {
    row_getter = [get, nrow, ncol](int64_t r, double* o) {
      for (int c = 0; c < ncol; ++c) o[c] = get(static_cast<int64_t>(c) * nrow + r);
    };
  }
  PredictRows(b, row_getter, nrow, ncol, predict_type, start_iteration, num_iteration,
              out_result, parameter);
  *out_len = static_cast<int64_t>(nrow) *
             b->NumPredictOneRow(start_iteration, num_iteration,
                                 predict_type == C_API_PREDICT_LEAF_INDEX,
                                 predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

int LGBM_BoosterPredictForMatSingleRow(BoosterHandle handle, const void* data, int data_type,
                                       int ncol, int is_row_major, int predict_type,
                                       int start_iteration, int num_iteration,
                                       const char* parameter, int64_t* out_len,
                                       double* out_result) {
  return LGBM_BoosterPredictForMat(handle, data, data_type, 1, ncol, is_row_major,
                                   predict_type, start_iteration, num_iteration, parameter,
                                   out_len, out_result);
}

int LGBM_BoosterPredictForCSR(BoosterHandle handle, const void* indptr, int indptr_type,
                              const int32_t* indices, const void* data, int data_type,
                              int64_t nindptr, int64_t nelem, int64_t num_col,
                              int predict_type, int start_iteration, int num_iteration,
                              const char*, int64_t* out_len, double* out_result) {
  API_BEGIN();
  (void)nelem;
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  auto ip = MakeIndptrGetter(indptr, indptr_type);
  auto val = MakeGetter(data, data_type);
  const int64_t nrow = nindptr - 1;
  auto row_getter = [&, ip, val, indices, num_col](int64_t r, double* o) {
    std::fill(o, o + num_col, 0.0);
    for (int64_t k = ip(r); k < ip(r + 1); ++k)
      if (indices[k] < num_col) o[indices[k]] = val(k);
  };
  PredictRows(b, row_getter, nrow, static_cast<int>(num_col), predict_type, start_iteration,
              num_iteration, out_result);
  *out_len = nrow * b->NumPredictOneRow(start_iteration, num_iteration,
                                        predict_type == C_API_PREDICT_LEAF_INDEX,
                                        predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

int LGBM_BoosterPredictForCSRSingleRow(BoosterHandle handle, const void* indptr,
                                       int indptr_type, const int32_t* indices,
                                       const void* data, int data_type, int64_t nindptr,
                                       int64_t nelem, int64_t num_col, int predict_type,
                                       int start_iteration, int num_iteration,
                                       const char* parameter, int64_t* out_len,
                                       double* out_result) {
  return LGBM_BoosterPredictForCSR(handle, indptr, indptr_type, indices, data, data_type,
                                   nindptr, nelem, num_col, predict_type, start_iteration,
                                   num_iteration, parameter, out_len, out_result);
}

int LGBM_BoosterPredictForCSC(BoosterHandle handle, const void* col_ptr, int col_ptr_type,
                              const int32_t* indices, const void* data, int data_type,
                              int64_t ncol_ptr, int64_t /*nelem*/, int64_t num_row,
                              int predict_type, int start_iteration, int num_iteration,
                              const char*, int64_t* out_len, double* out_result) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  auto cp = MakeIndptrGetter(col_ptr, col_ptr_type);
  auto val = MakeGetter(data, data_type);
  const int64_t num_col = ncol_ptr - 1;
  // transpose CSC -> dense row-major once (predict inputs are caller-bounded)
  std::vector<double> dense(static_cast<size_t>(num_row) * num_col, 0.0);
  for (int64_t c = 0; c < num_col; ++c)
    for (int64_t k = cp(c); k < cp(c + 1); ++k)
      if (indices[k] < num_row)
        dense[static_cast<size_t>(indices[k]) * num_col + c] = val(k);
  auto row_getter = [&dense, num_col](int64_t r, double* o) {
    memcpy(o, dense.data() + static_cast<size_t>(r) * num_col, sizeof(double) * num_col);
  };
  PredictRows(b, row_getter, num_row, static_cast<int>(num_col), predict_type,
              start_iteration, num_iteration, out_result);
  *out_len = num_row * b->NumPredictOneRow(start_iteration, num_iteration,
                                           predict_type == C_API_PREDICT_LEAF_INDEX,
                                           predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

int LGBM_BoosterPredictSparseOutput(BoosterHandle handle, const void* indptr,
                                    int indptr_type, const int32_t* indices,
                                    const void* data, int data_type, int64_t nindptr,
                                    int64_t /*nelem*/, int64_t num_col, int predict_type,
                                    int start_iteration, int num_iteration, const char*,
                                    int matrix_type, int64_t* out_len, void** out_indptr,
                                    int32_t** out_indices, void** out_data) {
  API_BEGIN();
  if (predict_type != C_API_PREDICT_CONTRIB)
    Log::Fatal("PredictSparseOutput only supports contribution (SHAP) prediction");
  if (matrix_type != 0)
    Log::Fatal("PredictSparseOutput currently emits CSR only (matrix_type=0)");
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  auto ip = MakeIndptrGetter(indptr, indptr_type);
  auto val = MakeGetter(data, data_type);
  const int64_t nrow = nindptr - 1;
  const int per_row = b->NumPredictOneRow(start_iteration, num_iteration, false, true);
  std::vector<double> dense(static_cast<size_t>(nrow) * per_row);
  auto row_getter = [&, indices, num_col](int64_t r, double* o) {
    std::fill(o, o + num_col, 0.0);
    for (int64_t k = ip(r); k < ip(r + 1); ++k)
      if (indices[k] < num_col) o[indices[k]] = val(k);
  };
  PredictRows(b, row_getter, nrow, static_cast<int>(num_col), predict_type,
              start_iteration, num_iteration, dense.data());
  // compact nonzeros into a fresh CSR triple owned by the caller
  std::vector<int64_t> rp(nrow + 1, 0);
  for (int64_t r = 0; r < nrow; ++r) {
    int64_t nz = 0;
    for (int j = 0; j < per_row; ++j)
      if (dense[static_cast<size_t>(r) * per_row + j] != 0.0) ++nz;
    rp[r + 1] = rp[r] + nz;
  }
  const int64_t total = rp[nrow];
  int64_t* o_indptr = new int64_t[nrow + 1];
  int32_t* o_indices = new int32_t[std::max<int64_t>(total, 1)];
  double* o_data = new double[std::max<int64_t>(total, 1)];
  memcpy(o_indptr, rp.data(), sizeof(int64_t) * (nrow + 1));
  for (int64_t r = 0; r < nrow; ++r) {
    int64_t w = rp[r];
    for (int j = 0; j < per_row; ++j) {
      const double v = dense[static_cast<size_t>(r) * per_row + j];
      if (v != 0.0) {
        o_indices[w] = j;
        o_data[w] = v;
        ++w;
      }
    }
  }
  *out_len = total;
  *out_indptr = o_indptr;
  *out_indices = o_indices;
  *out_data = o_data;
  API_END();
}

int LGBM_BoosterFreePredictSparse(void* indptr, int32_t* indices, void* data,
                                  int /*indptr_type*/, int /*data_type*/) {
  API_BEGIN();
  delete[] static_cast<int64_t*>(indptr);
  delete[] indices;
  delete[] static_cast<double*>(data);
  API_END();
}

int LGBM_BoosterPredictForArrow(BoosterHandle handle, int64_t n_chunks,
                                const struct ArrowArray* chunks,
                                const struct ArrowSchema* schema, int predict_type,
                                int start_iteration, int num_iteration, const char*,
                                int64_t* out_len, double* out_result) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  MIGBM_CHECK(schema->n_children > 0);
  const int ncol = static_cast<int>(schema->n_children);
  std::vector<int64_t> chunk_start(n_chunks + 1, 0);
  std::vector<std::vector<std::function<double(int64_t)>>> getters(n_chunks);
  for (int64_t c = 0; c < n_chunks; ++c) {
    chunk_start[c + 1] = chunk_start[c] + chunks[c].length;
    getters[c].resize(ncol);
    for (int f = 0; f < ncol; ++f)
      getters[c][f] = ArrowColGetter(chunks[c].children[f], schema->children[f]->format);
  }
  const int64_t nrow = chunk_start[n_chunks];
  auto row_getter = [&](int64_t r, double* o) {
    int64_t lo = 0, hi = n_chunks - 1;
    while (lo < hi) {
      const int64_t mid = (lo + hi + 1) >> 1;
      if (r >= chunk_start[mid]) lo = mid;
      else hi = mid - 1;
    }
    for (int f = 0; f < ncol; ++f) o[f] = getters[lo][f](r - chunk_start[lo]);
  };
  PredictRows(b, row_getter, nrow, ncol, predict_type, start_iteration, num_iteration,
              out_result);
  *out_len = nrow * b->NumPredictOneRow(start_iteration, num_iteration,
                                        predict_type == C_API_PREDICT_LEAF_INDEX,
                                        predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

int LGBM_BoosterPredictForMats(BoosterHandle handle, const void** data, int data_type,
                               int32_t nrow, int32_t ncol, int predict_type,
                               int start_iteration, int num_iteration, const char*,
                               int64_t* out_len, double* out_result) {
  API_BEGIN();
  auto* b = static_cast<BoosterWrapper*>(handle)->boosting();
  auto row_getter = [data, data_type, ncol](int64_t r, double* o) {
    auto get = MakeGetter(data[r], data_type);
    for (int c = 0; c < ncol; ++c) o[c] = get(c);
  };
  PredictRows(b, row_getter, nrow, ncol, predict_type, start_iteration, num_iteration,
              out_result);
  *out_len = static_cast<int64_t>(nrow) *
             b->NumPredictOneRow(start_iteration, num_iteration,
                                 predict_type == C_API_PREDICT_LEAF_INDEX,
                                 predict_type == C_API_PREDICT_CONTRIB);
  API_END();
}

/*! bound single-row predict config (reference FastConfig semantics: parameters and
 *  shape parsed once at Init; per-row calls do no setup work) */
struct FastConfig {
  BoosterHandle booster;
  int predict_type, start_iteration, num_iteration, data_type;
  int64_t ncol;
};

int LGBM_BoosterPredictForMatSingleRowFastInit(BoosterHandle handle, int predict_type,
                                               int start_iteration, int num_iteration,
                                               int data_type, int32_t ncol, const char*,
                                               FastConfigHandle* out_fastConfig) {
  API_BEGIN();
  *out_fastConfig = new FastConfig{handle, predict_type, start_iteration, num_iteration,
                                   data_type, ncol};
  API_END();
}

int LGBM_BoosterPredictForMatSingleRowFast(FastConfigHandle fastConfig_handle,
                                           const void* data, int64_t* out_len,
                                           double* out_result) {
  auto* fc = static_cast<FastConfig*>(fastConfig_handle);
  return LGBM_BoosterPredictForMat(fc->booster, data, fc->data_type, 1,
                                   static_cast<int32_t>(fc->ncol), 1, fc->predict_type,
                                   fc->start_iteration, fc->num_iteration, "", out_len,
                                   out_result);
}

int LGBM_BoosterPredictForCSRSingleRowFastInit(BoosterHandle handle, int predict_type,
                                               int start_iteration, int num_iteration,
                                               int data_type, int64_t num_col, const char*,
                                               FastConfigHandle* out_fastConfig) {
  API_BEGIN();
  *out_fastConfig = new FastConfig{handle, predict_type, start_iteration, num_iteration,
                                   data_type, num_col};
  API_END();
}

int LGBM_BoosterPredictForCSRSingleRowFast(FastConfigHandle fastConfig_handle,
                                           const void* indptr, int indptr_type,
                                           const int32_t* indices, const void* data,
                                           int64_t nindptr, int64_t nelem, int64_t* out_len,
                                           double* out_result) {
  auto* fc = static_cast<FastConfig*>(fastConfig_handle);
  return LGBM_BoosterPredictForCSR(fc->booster, indptr, indptr_type, indices, data,
                                   fc->data_type, nindptr, nelem, fc->ncol, fc->predict_type,
                                   fc->start_iteration, fc->num_iteration, "", out_len,
                                   out_result);
}

int LGBM_FastConfigFree(FastConfigHandle fastConfig) {
  API_BEGIN();
  delete static_cast<FastConfig*>(fastConfig);
  API_END();
}

int LGBM_BoosterPredictForFile(BoosterHandle handle, const char* data_filename,
                               int data_has_header, int predict_type, int start_iteration,
                               int num_iteration, const char* parameter,
                               const char* result_filename) {
  API_BEGIN();
  auto* bw = static_cast<BoosterWrapper*>(handle);
  auto* b = bw->boosting();
  Config cfg;
  cfg.Set(Config::Str2Map(parameter));
  cfg.header = data_has_header != 0;
  // load raw rows (label column dropped if present)
  auto rows = migbm::LoadRawRowsForPredict(data_filename, cfg, b->MaxFeatureIdx() + 1);
  const int64_t nrow = static_cast<int64_t>(rows.size());
  const int ncol = b->MaxFeatureIdx() + 1;
  const int per_row = b->NumPredictOneRow(start_iteration, num_iteration,
                                          predict_type == C_API_PREDICT_LEAF_INDEX,
                                          predict_type == C_API_PREDICT_CONTRIB);
  std::vector<double> out(nrow * per_row);
  auto row_getter = [&](int64_t r, double* o) {
    for (int c = 0; c < ncol; ++c)
      o[c] = c < static_cast<int>(rows[r].size()) ? rows[r][c] : 0.0;
  };
  PredictRows(b, row_getter, nrow, ncol, predict_type, start_iteration, num_iteration,
              out.data());
  FILE* fp = fopen(result_filename, "w");
  if (!fp) Log::Fatal("Cannot open %s", result_filename);
  for (int64_t i = 0; i < nrow; ++i) {
    for (int k = 0; k < per_row; ++k) {
      if (k) fputc('\t', fp);
      fprintf(fp, "%.17g", out[i * per_row + k]);
    }
    fputc('\n', fp);
  }
  fclose(fp);
  API_END();
}

int LGBM_BoosterSaveModel(BoosterHandle handle, int start_iteration, int num_iteration,
                          int feature_importance_type, const char* filename) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->boosting()->SaveModelToFile(
      start_iteration, num_iteration, feature_importance_type, filename);
  API_END();
}

int LGBM_BoosterSaveModelToString(BoosterHandle handle, int start_iteration, int num_iteration,
                                  int feature_importance_type, int64_t buffer_len,
                                  int64_t* out_len, char* out_str) {
  API_BEGIN();
  auto s = static_cast<BoosterWrapper*>(handle)->boosting()->SaveModelToString(
      start_iteration, num_iteration, feature_importance_type);
  CopyToBuffer(s, buffer_len, out_len, out_str);
  API_END();
}

int LGBM_BoosterDumpModel(BoosterHandle handle, int start_iteration, int num_iteration,
                          int feature_importance_type, int64_t buffer_len, int64_t* out_len,
                          char* out_str) {
  API_BEGIN();
  auto s = static_cast<BoosterWrapper*>(handle)->boosting()->DumpModel(
      start_iteration, num_iteration, feature_importance_type);
  CopyToBuffer(s, buffer_len, out_len, out_str);
  API_END();
}

int LGBM_BoosterGetLeafValue(BoosterHandle handle, int tree_idx, int leaf_idx,
                             double* out_val) {
  API_BEGIN();
  *out_val = static_cast<BoosterWrapper*>(handle)->boosting()->GetLeafValue(tree_idx, leaf_idx);
  API_END();
}

int LGBM_BoosterSetLeafValue(BoosterHandle handle, int tree_idx, int leaf_idx, double val) {
  API_BEGIN();
  static_cast<BoosterWrapper*>(handle)->boosting()->SetLeafValue(tree_idx, leaf_idx, val);
  API_END();
}

int LGBM_BoosterFeatureImportance(BoosterHandle handle, int num_iteration, int importance_type,
                                  double* out_results) {
  API_BEGIN();
  auto imp = static_cast<BoosterWrapper*>(handle)->boosting()->FeatureImportance(
      num_iteration, importance_type);
  std::copy(imp.begin(), imp.end(), out_results);
  API_END();
}

int LGBM_BoosterGetUpperBoundValue(BoosterHandle handle, double* out) {
  API_BEGIN();
  *out = static_cast<BoosterWrapper*>(handle)->boosting()->GetUpperBoundValue();
  API_END();
}

int LGBM_BoosterGetLowerBoundValue(BoosterHandle handle, double* out) {
  API_BEGIN();
  *out = static_cast<BoosterWrapper*>(handle)->boosting()->GetLowerBoundValue();
  API_END();
}

int LGBM_BoosterGetLinear(BoosterHandle handle, int* out) {
  API_BEGIN();
  *out = static_cast<BoosterWrapper*>(handle)->boosting()->IsLinear() ? 1 : 0;
  API_END();
}

// ================================================================== network
namespace {
AllgatherExtFunction g_allgather_ext = nullptr;
void AllgatherBridge(const char* input, int input_size, char* output) {
  // uniform block layout
  const int world = Network::num_machines();
  std::vector<int> starts(world), lens(world);
  for (int i = 0; i < world; ++i) {
    starts[i] = i * input_size;
    lens[i] = input_size;
  }
  g_allgather_ext(const_cast<char*>(input), input_size, starts.data(), lens.data(), world,
                  output, input_size * world);
}
}  // namespace

int LGBM_NetworkInit(const char* machines, int local_listen_port, int listen_time_out,
                     int num_machines) {
  API_BEGIN();
  NetworkInitSockets(machines ? machines : "", local_listen_port, listen_time_out,
                     num_machines);
  API_END();
}

int LGBM_NetworkFree() {
  API_BEGIN();
  Network::Free();
  NetworkFreeSockets();
  g_allgather_ext = nullptr;
  API_END();
}

int LGBM_NetworkInitWithFunctions(int num_machines, int rank, void* /*reduce_scatter_ext_fun*/,
                                  void* allgather_ext_fun) {
  API_BEGIN();
  g_allgather_ext = reinterpret_cast<AllgatherExtFunction>(allgather_ext_fun);
  Network::Init(num_machines, rank, AllgatherBridge);
  API_END();
}
