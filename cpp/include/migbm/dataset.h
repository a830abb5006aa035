/*!
 * migbm Dataset — binned training data + Metadata (label/weight/query/init_score).
 * Capability parity target: reference include/LightGBM/dataset.h, src/io/dataset.cpp,
 * src/io/metadata.cpp, src/io/dataset_loader.cpp. Fresh design: dense per-feature bin
 * columns (uint8/uint16) on the host, plus an optional row-major packed matrix view used
 * by the HIP learner (built once, resident in HBM).
 */
#ifndef MIGBM_DATASET_H_
#define MIGBM_DATASET_H_

#include "bin.h"
#include "common.h"
#include "config.h"

#include <memory>
#include <string>
#include <algorithm>
#include <vector>

namespace migbm {

/*! Per-row ancillary data. Parity: reference Metadata (dataset.h:48-400). */
class Metadata {
 public:
  void Init(data_size_t num_data, bool has_weight, bool has_query);
  void SetLabel(const float* label, data_size_t n);
  void SetWeights(const float* w, data_size_t n);
  void SetQuery(const int32_t* group_sizes, data_size_t n_groups);
  void SetQueryBoundaries(std::vector<data_size_t> boundaries);
  void SetInitScore(const double* s, int64_t n);
  void SetPosition(const int32_t* p, data_size_t n);

  data_size_t num_data() const { return num_data_; }
  const label_t* label() const { return label_.data(); }
  const label_t* weights() const { return weights_.empty() ? nullptr : weights_.data(); }
  const data_size_t* query_boundaries() const {
    return query_boundaries_.empty() ? nullptr : query_boundaries_.data();
  }
  data_size_t num_queries() const {
    return query_boundaries_.empty() ? 0 : static_cast<data_size_t>(query_boundaries_.size() - 1);
  }
  const label_t* query_weights() const {
    return query_weights_.empty() ? nullptr : query_weights_.data();
  }
  const double* init_score() const { return init_score_.empty() ? nullptr : init_score_.data(); }
  int64_t num_init_score() const { return static_cast<int64_t>(init_score_.size()); }
  const int32_t* positions() const { return positions_.empty() ? nullptr : positions_.data(); }

  std::vector<label_t>& mutable_label() { return label_; }

  /*! streaming-push metadata (reference PushRows*WithMetadata semantics): values
   *  land positionally; query ids are run-length encoded into boundaries by
   *  FinalizeStreamedQueries (called from LGBM_DatasetMarkFinished). */
  void SetStreamedWeights(int64_t start, const float* w, int32_t n) {
    if (weights_.size() != static_cast<size_t>(num_data_)) weights_.assign(num_data_, 1.0f);
    for (int32_t i = 0; i < n; ++i) weights_[start + i] = w[i];
  }
  void SetStreamedInitScores(int64_t start, const double* s, int32_t n) {
    if (init_score_.size() < static_cast<size_t>(num_data_))
      init_score_.assign(num_data_, 0.0);
    for (int32_t i = 0; i < n; ++i) init_score_[start + i] = s[i];
  }
  void SetStreamedQueryIds(int64_t start, const int32_t* q, int32_t n) {
    if (streamed_query_ids_.size() != static_cast<size_t>(num_data_))
      streamed_query_ids_.assign(num_data_, 0);
    for (int32_t i = 0; i < n; ++i) streamed_query_ids_[start + i] = q[i];
  }
  void FinalizeStreamedQueries() {
    if (streamed_query_ids_.empty()) return;
    query_boundaries_.clear();
    query_boundaries_.push_back(0);
    for (data_size_t i = 1; i < num_data_; ++i)
      if (streamed_query_ids_[i] != streamed_query_ids_[i - 1]) query_boundaries_.push_back(i);
    query_boundaries_.push_back(num_data_);
    streamed_query_ids_.clear();
  }

 private:
  data_size_t num_data_ = 0;
  std::vector<label_t> label_;
  std::vector<label_t> weights_;
  std::vector<data_size_t> query_boundaries_;
  std::vector<label_t> query_weights_;
  std::vector<double> init_score_;
  std::vector<int32_t> positions_;
  std::vector<int32_t> streamed_query_ids_;
};

/*! Bin column: dense (uint8 when num_bin<=256 else uint16) or sparse.
 *  Sparse mode stores (sorted row, bin) pairs for rows whose bin differs from the
 *  column's default bin — chosen at construction for high-default-fraction
 *  features (capability parity: reference SparseBin / is_enable_sparse; fresh
 *  row-list design instead of delta encoding). */
class BinColumn {
 public:
  void Init(data_size_t n, int num_bin) {
    is_sparse_ = false;
    is4_ = false;
    is16_ = num_bin > 256;
    if (is16_) d16_.assign(n, 0);
    else d8_.assign(n, 0);
  }
  /*! 4-bit packed dense column (2 bins per byte) for num_bin <= 16
   *  (capability parity: reference dense_bin.hpp IS_4BIT). */
  void Init4(data_size_t n) {
    is_sparse_ = false;
    is16_ = false;
    is4_ = true;
    d8_.assign((static_cast<size_t>(n) + 1) / 2, 0);
  }
  void InitSparse(data_size_t n, int num_bin, uint32_t default_bin) {
    is_sparse_ = true;
    is16_ = num_bin > 256;
    num_data_ = n;
    default_bin_ = default_bin;
    s_rows_.clear();
    s_bins_.clear();
  }
  /*! dense write, or sparse append (sparse: rows must arrive in ascending order) */
  inline void Set(data_size_t i, uint32_t b) {
    if (is_sparse_) {
      if (b == default_bin_) return;
      s_rows_.push_back(i);
      s_bins_.push_back(b);
      return;
    }
    if (is4_) {
      uint8_t& byte = d8_[i >> 1];
      const int sh = (i & 1) * 4;
      byte = static_cast<uint8_t>((byte & ~(0xF << sh)) | ((b & 0xF) << sh));
      return;
    }
    if (is16_) d16_[i] = static_cast<uint16_t>(b);
    else d8_[i] = static_cast<uint8_t>(b);
  }
  inline uint32_t Get(data_size_t i) const {
    if (is_sparse_) {
      auto it = std::lower_bound(s_rows_.begin(), s_rows_.end(), i);
      if (it != s_rows_.end() && *it == i) return s_bins_[it - s_rows_.begin()];
      return default_bin_;
    }
    if (is4_) return (d8_[i >> 1] >> ((i & 1) * 4)) & 0xF;
    return is16_ ? d16_[i] : d8_[i];
  }
  bool is4() const { return is4_; }
  bool is16() const { return is16_; }
  bool is_sparse() const { return is_sparse_; }
  uint32_t default_bin() const { return default_bin_; }
  data_size_t nnz() const { return static_cast<data_size_t>(s_rows_.size()); }
  const data_size_t* sparse_rows() const { return s_rows_.data(); }
  const uint16_t* sparse_bins() const { return s_bins_.data(); }
  const uint8_t* data8() const { return d8_.data(); }
  const uint16_t* data16() const { return d16_.data(); }
  std::vector<data_size_t>& mutable_sparse_rows() { return s_rows_; }
  std::vector<uint16_t>& mutable_sparse_bins() { return s_bins_; }

 private:
  bool is16_ = false;
  bool is4_ = false;
  bool is_sparse_ = false;
  uint32_t default_bin_ = 0;
  data_size_t num_data_ = 0;
  std::vector<uint8_t> d8_;
  std::vector<uint16_t> d16_;
  std::vector<data_size_t> s_rows_;   // ascending
  std::vector<uint16_t> s_bins_;
};

class Dataset {
 public:
  Dataset() = default;
  explicit Dataset(data_size_t num_data) : num_data_(num_data) {}

  /*! Build bin mappers + columns from a dense matrix.
   *  \param sample_getter value at (row, col); NaN allowed
   *  \param categorical per-original-column flag */
  void ConstructFromMat(const std::function<double(data_size_t, int)>& get, data_size_t nrow,
                        int ncol, const Config& cfg, const std::vector<int8_t>& categorical);
  /*! single-rank body (no distributed bin-mapper sync) */
  void ConstructFromMatLocal(const std::function<double(data_size_t, int)>& get,
                             data_size_t nrow, int ncol, const Config& cfg,
                             const std::vector<int8_t>& categorical);

  /*! Build an aligned valid set re-using this (train) dataset's bin mappers. */
  std::unique_ptr<Dataset> CreateValid(const std::function<double(data_size_t, int)>& get,
                                       data_size_t nrow) const;

  /*! True when `other`'s per-feature bin mappers match this dataset's — required
   *  before scoring trees trained on `other`'s bins against this data.
   *  Reference parity: DatasetLoader::CheckAlignWithOtherDataset semantics. */
  bool AlignsWith(const Dataset* other) const;

  /*! Column-wise histogram build over an ordered index subset.
   *  hist layout: per used feature f at hist_offset(f)*2, (sum_grad, sum_hess) pairs.
   *  ordered_grad/hess must be pre-gathered to match data_indices order. */
  void ConstructHistograms(const std::vector<int8_t>& is_feature_used,
                           const data_size_t* data_indices, data_size_t num_data,
                           const score_t* ordered_grad, const score_t* ordered_hess,
                           hist_t* hist) const {
    ConstructHistograms(is_feature_used, data_indices, num_data, ordered_grad, ordered_hess,
                        nullptr, nullptr, nullptr, hist);
  }
  /*! Full form: row_grad/row_hess are the FULL per-row gradient arrays and
   *  in_leaf is a per-row membership byte mask — both required only when the
   *  dataset has sparse columns and the leaf is a strict subset (the sparse path
   *  iterates column nonzeros, which are row- not position-indexed). */
  void ConstructHistograms(const std::vector<int8_t>& is_feature_used,
                           const data_size_t* data_indices, data_size_t num_data,
                           const score_t* ordered_grad, const score_t* ordered_hess,
                           const score_t* row_grad, const score_t* row_hess,
                           const uint8_t* in_leaf, hist_t* hist) const;
  bool has_sparse() const { return has_sparse_; }
  /*! Row-wise histogram over the packed row-major view: one pass over the rows,
   *  per-thread private histograms merged at the end (the migbm analogue of the
   *  reference's row-wise MultiValBin mode; selection is empirical, see
   *  SerialTreeLearner::ComputeHistogram). Covers every feature. */
  void ConstructHistogramsRowWise(const data_size_t* data_indices, data_size_t num_data,
                                  const score_t* ordered_grad,
                                  const score_t* ordered_hess, hist_t* hist) const {
    ConstructHistogramsRowWise(data_indices, num_data, ordered_grad, ordered_hess, hist,
                               false);
  }
  /*! row_indexed=true: grad/hess arrays are indexed by ROW id (the full per-row
   *  arrays) — skips the ordered-gather pass entirely. */
  void ConstructHistogramsRowWise(const data_size_t* data_indices, data_size_t num_data,
                                  const score_t* grad, const score_t* hess, hist_t* hist,
                                  bool row_indexed) const;
  /*! non-constant-hessian fast path: gh = interleaved (grad, hess) pairs indexed
   *  by row id (one 8B load per row instead of two 4B gathers from two arrays);
   *  4-row interleave for store-to-load latency hiding. */
  void ConstructHistogramsRowWiseGH(const data_size_t* data_indices, data_size_t num_data,
                                    const score_t* gh, hist_t* hist) const;
  /*! default bin of a sparse feature (the bin MaterializeDefaultBins must
   *  reconstruct from leaf totals), or -1 for dense features. */
  int feature_sparse_default_bin(int f) const {
    const BinColumn& c = columns_[col_of_feature_[f]];
    return c.is_sparse() ? static_cast<int>(c.default_bin()) : -1;
  }

  /*! Histogram for one inner feature into out[2*bin]. */
  void ConstructHistogramForFeature(int fidx, const data_size_t* data_indices,
                                    data_size_t num_data, const score_t* ordered_grad,
                                    const score_t* ordered_hess, hist_t* out) const;

  data_size_t num_data() const { return num_data_; }
  int num_features() const { return static_cast<int>(bin_mappers_.size()); }
  int num_total_features() const { return num_total_features_; }
  int InnerFeatureIndex(int orig) const { return used_feature_map_[orig]; }
  int RealFeatureIndex(int inner) const { return real_feature_index_[inner]; }
  const BinMapper* FeatureBinMapper(int inner) const { return bin_mappers_[inner].get(); }
  int FeatureNumBin(int inner) const { return bin_mappers_[inner]->num_bin(); }
  uint32_t hist_offset(int inner) const { return hist_offsets_[inner]; }
  int num_total_bin() const { return num_total_bin_; }
  /*! feature-level bin for a row, decoding EFB bundle columns when present */
  inline uint32_t GetBin(data_size_t row, int inner) const {
    const uint32_t v = columns_[col_of_feature_[inner]].Get(row);
    if (!feature_bundled_[inner]) return v;
    const uint32_t off = off_in_col_[inner];
    const uint32_t span = static_cast<uint32_t>(bin_mappers_[inner]->num_bin()) - 1;
    return (v >= off && v < off + span) ? v - off + 1 : 0;
  }
  int num_columns() const { return static_cast<int>(columns_.size()); }
  int feature_column(int inner) const { return col_of_feature_[inner]; }
  bool feature_bundled(int inner) const { return feature_bundled_[inner] != 0; }
  bool has_bundles() const { return has_bundles_; }
  /*! features stored in column c (one entry for unbundled columns) */
  const std::vector<int>& column_features(int c) const { return column_features_[c]; }
  const BinColumn& column(int c) const { return columns_[c]; }

  Metadata& metadata() { return metadata_; }
  const Metadata& metadata() const { return metadata_; }

  const std::vector<std::string>& feature_names() const { return feature_names_; }
  void set_feature_names(const std::vector<std::string>& names);
  std::string FeatureInfoString() const;  // "feature_infos=" payload
  std::string GetFeatureName(int orig) const {
    return orig < static_cast<int>(feature_names_.size()) ? feature_names_[orig]
                                                          : "Column_" + std::to_string(orig);
  }

  /*! Threshold raw value for (inner feature, bin): upper bound of the bin. */
  double RealThreshold(int inner, uint32_t bin) const {
    return bin_mappers_[inner]->BinToValue(bin);
  }

  /*! Row-major packed view for the HIP learner: one byte (or 2) per used feature,
   *  padded row stride. Built lazily. */
  struct RowMajorView {
    std::vector<uint8_t> data;     // uint8 path (all features <=256 bins incl. nan bin)
    std::vector<uint16_t> data16;  // fallback
    int row_stride = 0;            // elements per row (padded)
    bool is16 = false;
  };
  const RowMajorView& GetRowMajorView() const;

  /*! Used by tests / CLI: binary serialization of the dataset. */
  void SaveBinaryFile(const char* filename) const;
  static std::unique_ptr<Dataset> LoadFromBinFile(const char* filename);
  static bool IsBinFile(const char* filename);

  void DumpTextFile(const char* filename) const;

  /*! Subset copy of rows (bagging subset path / GetSubset C API). */
  std::unique_ptr<Dataset> Subset(const data_size_t* indices, data_size_t n) const;

  /*! Push a single raw row (streaming API). Requires mappers already built. */
  void PushRawRow(data_size_t row, const double* values, int ncol);

  /*! Build bin mappers from per-column sampled values (streaming bootstrap).
   *  Parity: reference DatasetLoader::ConstructFromSampleData. */
  void ConstructFromSampleData(double** sample_values, int** sample_indices, int ncol,
                               const int* num_per_col, int num_sample_row,
                               data_size_t num_local_row, const Config& cfg,
                               const std::vector<int8_t>& categorical);
  /*! Clone this dataset's mappers into an empty dataset with num_rows rows. */
  std::unique_ptr<Dataset> CreateByReference(data_size_t num_rows) const;
  /*! Serialize mappers+schema (no row data) for cross-process reference sharing. */
  std::string SerializeReference() const;
  static std::unique_ptr<Dataset> FromSerializedReference(const char* buf, size_t len,
                                                          data_size_t num_rows);

  std::vector<int8_t> categorical_flags_;   // per original column
  bool has_raw() const { return !raw_values_.empty(); }
  float raw_value(int inner, data_size_t row) const { return raw_values_[inner][row]; }
  const float* raw_column(int inner) const { return raw_values_[inner].data(); }

 private:
  friend class DatasetLoader;
  void FinishBinMappers(const Config& cfg);
  /*! EFB: greedily bundle sparse mutually-exclusive features into shared columns.
   *  Fills col_of_feature_/off_in_col_/column_features_ (identity when disabled).
   *  Parity: reference Dataset::FindGroups/FastFeatureBundling (re-derived). */
  void PlanBundles(const Config& cfg, const std::function<double(data_size_t, int)>& get,
                   const std::vector<data_size_t>& sample_idx);
  void CopyBundlingFrom(const Dataset& other) {
    col_of_feature_ = other.col_of_feature_;
    off_in_col_ = other.off_in_col_;
    feature_bundled_ = other.feature_bundled_;
    column_features_ = other.column_features_;
    has_bundles_ = other.has_bundles_;
    has_sparse_ = other.has_sparse_;
  }

  data_size_t num_data_ = 0;
  int num_total_features_ = 0;

 public:
  /*! Append every feature (and its storage columns) of `other` to this dataset.
   *  Same row count required. Reference parity: LGBM_DatasetAddFeaturesFrom /
   *  Dataset::AddFeaturesFrom (src/io/dataset.cpp). */
  void AddFeaturesFrom(const Dataset* other);

 private:
  std::vector<int> used_feature_map_;        // orig -> inner (-1 trivial)
  std::vector<int> real_feature_index_;      // inner -> orig
  std::vector<std::unique_ptr<BinMapper>> bin_mappers_;  // per inner feature
  // storage columns: one per unbundled feature + one per EFB bundle
  std::vector<BinColumn> columns_;
  std::vector<int> col_of_feature_;          // inner feature -> column
  std::vector<uint32_t> off_in_col_;         // bundle offset (bundled features)
  std::vector<int8_t> feature_bundled_;
  std::vector<std::vector<int>> column_features_;
  bool has_bundles_ = false;
  bool has_sparse_ = false;
  std::vector<uint32_t> hist_offsets_;
  int num_total_bin_ = 0;
  Metadata metadata_;
  std::vector<std::vector<float>> raw_values_;  // per inner feature (linear_tree only)
  std::vector<std::string> feature_names_;
  mutable RowMajorView row_view_;
  mutable bool row_view_built_ = false;
};

/*! Text-file loading (CSV/TSV/LibSVM autodetect).
 *  Parity: reference src/io/dataset_loader.cpp + parser.cpp. */
class DatasetLoader {
 public:
  DatasetLoader(const Config& cfg, int label_idx = 0) : cfg_(cfg), label_idx_(label_idx) {}
  std::unique_ptr<Dataset> LoadFromFile(const char* filename, int rank = 0, int num_machines = 1);
  std::unique_ptr<Dataset> LoadFromFileAlignWithOtherDataset(const char* filename,
                                                             const Dataset* train);

 protected:
  void ParseFile(const char* filename, std::vector<std::vector<double>>* rows,
                 std::vector<float>* labels, std::vector<float>* weights,
                 std::vector<int32_t>* groups, int* ncol, int rank, int num_machines);
  Config cfg_;
  int label_idx_;
};

/*! Raw-row loading used by LGBM_BoosterPredictForFile. */
std::vector<std::vector<double>> LoadRawRowsForPredict(const char* filename, const Config& cfg,
                                                       int expected_ncol);

}  // namespace migbm

#endif  // MIGBM_DATASET_H_
