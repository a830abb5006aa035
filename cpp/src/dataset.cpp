/*! migbm Dataset implementation: sampling + binning + dense bin columns + histograms +
 *  row-major packed view for the HIP learner + binary save/load.
 *  Parity target: reference src/io/dataset.cpp (ConstructHistogramsInner, SaveBinaryFile),
 *  src/io/metadata.cpp. Fresh implementation. */
#include "migbm/dataset.h"
#include "migbm/network.h"

#include <cstdio>
#include <numeric>

namespace migbm {

// ------------------------------------------------------------------ Metadata
void Metadata::Init(data_size_t num_data, bool has_weight, bool has_query) {
  num_data_ = num_data;
  label_.assign(num_data, 0.0f);
  if (has_weight) weights_.assign(num_data, 1.0f);
  if (has_query) query_boundaries_.clear();
}

void Metadata::SetLabel(const float* label, data_size_t n) {
  for (data_size_t i = 0; i < n; ++i) {
    if (!std::isfinite(label[i]))
      Log::Fatal("Label at index %d is not finite (%f); labels must not contain NaN/Inf",
                 i, static_cast<double>(label[i]));
  }
  num_data_ = n;
  label_.assign(label, label + n);
}

void Metadata::SetWeights(const float* w, data_size_t n) {
  if (w == nullptr || n == 0) { weights_.clear(); return; }
  MIGBM_CHECK_EQ(n, num_data_);
  weights_.assign(w, w + n);
}

void Metadata::SetQuery(const int32_t* group_sizes, data_size_t n_groups) {
  if (group_sizes == nullptr || n_groups == 0) { query_boundaries_.clear(); return; }
  query_boundaries_.resize(n_groups + 1);
  query_boundaries_[0] = 0;
  for (data_size_t i = 0; i < n_groups; ++i)
    query_boundaries_[i + 1] = query_boundaries_[i] + group_sizes[i];
  MIGBM_CHECK_EQ(query_boundaries_.back(), num_data_);
}

void Metadata::SetQueryBoundaries(std::vector<data_size_t> boundaries) {
  query_boundaries_ = std::move(boundaries);
}

void Metadata::SetInitScore(const double* s, int64_t n) {
  if (s == nullptr || n == 0) { init_score_.clear(); return; }
  init_score_.assign(s, s + n);
}

void Metadata::SetPosition(const int32_t* p, data_size_t n) {
  if (p == nullptr || n == 0) { positions_.clear(); return; }
  MIGBM_CHECK_EQ(n, num_data_);
  positions_.assign(p, p + n);
}

// ------------------------------------------------------------------ Dataset
void Dataset::FinishBinMappers(const Config&) {
  hist_offsets_.resize(bin_mappers_.size());
  uint32_t off = 0;
  for (size_t i = 0; i < bin_mappers_.size(); ++i) {
    hist_offsets_[i] = off;
    off += bin_mappers_[i]->num_bin();
  }
  num_total_bin_ = static_cast<int>(off);
}

/*! parse forcedbins_filename JSON: [{"feature": i, "bin_upper_bound": [v, ...]}, ...]
 *  (reference DatasetLoader forced-bins format). */
static std::unordered_map<int, std::vector<double>> ParseForcedBins(const std::string& path) {
  std::unordered_map<int, std::vector<double>> out;
  if (path.empty()) return out;
  FILE* fp = fopen(path.c_str(), "rb");
  if (!fp) {
    Log::Warning("Cannot open forced bins file %s", path.c_str());
    return out;
  }
  std::string content;
  char buf[4096];
  size_t got;
  while ((got = fread(buf, 1, sizeof(buf), fp)) > 0) content.append(buf, got);
  fclose(fp);
  size_t pos = 0;
  auto skip = [&]() {
    while (pos < content.size() &&
           (isspace(static_cast<unsigned char>(content[pos])) || content[pos] == ','))
      ++pos;
  };
  while (true) {
    size_t obj = content.find('{', pos);
    if (obj == std::string::npos) break;
    size_t end = content.find('}', obj);
    if (end == std::string::npos) break;
    std::string o = content.substr(obj, end - obj + 1);
    int feat = -1;
    std::vector<double> bounds;
    size_t fpos = o.find("\"feature\"");
    if (fpos != std::string::npos) {
      fpos = o.find(':', fpos);
      if (fpos != std::string::npos) feat = atoi(o.c_str() + fpos + 1);
    }
    size_t bpos = o.find("\"bin_upper_bound\"");
    if (bpos != std::string::npos) {
      size_t lb = o.find('[', bpos);
      size_t rb = o.find(']', bpos);
      if (lb != std::string::npos && rb != std::string::npos) {
        std::string arr = o.substr(lb + 1, rb - lb - 1);
        for (auto& tok : Common::Split(arr.c_str(), ',')) {
          auto t = Common::Trim(tok);
          if (!t.empty()) bounds.push_back(atof(t.c_str()));
        }
      }
    }
    if (feat >= 0 && !bounds.empty()) out[feat] = std::move(bounds);
    pos = end + 1;
    skip();
  }
  return out;
}

void Dataset::ConstructFromMat(const std::function<double(data_size_t, int)>& get,
                               data_size_t nrow, int ncol, const Config& cfg,
                               const std::vector<int8_t>& categorical) {
  if (Network::is_distributed()) {
    // distributed in-memory construction: every rank must end with IDENTICAL
    // bin mappers. Gather a per-rank row sample into a union sample matrix
    // (same bytes everywhere), find bins on it, then bin the local shard
    // (reference ConstructBinMappersFromTextData sample-sync parity).
    const int world = Network::num_machines();
    const data_size_t cap =
        std::max<data_size_t>(1, cfg.bin_construct_sample_cnt / std::max(1, world));
    const data_size_t s = std::min(nrow, cap);
    std::vector<double> flat(static_cast<size_t>(s) * ncol);
    const double stride = s > 0 ? static_cast<double>(nrow) / s : 1.0;
    for (data_size_t i = 0; i < s; ++i) {
      const data_size_t r =
          std::min<data_size_t>(nrow - 1, static_cast<data_size_t>(i * stride));
      for (int c = 0; c < ncol; ++c) flat[static_cast<size_t>(i) * ncol + c] = get(r, c);
    }
    std::vector<int64_t> counts(world, 0);
    int64_t mine = s;
    Network::Allgather(reinterpret_cast<const char*>(&mine), sizeof(int64_t),
                       reinterpret_cast<char*>(counts.data()));
    int64_t total_s = 0;
    std::vector<int> sizes(world);
    for (int r = 0; r < world; ++r) {
      total_s += counts[r];
      sizes[r] = static_cast<int>(counts[r] * ncol * sizeof(double));
    }
    std::vector<double> uni(static_cast<size_t>(total_s) * ncol);
    Network::AllgatherV(reinterpret_cast<const char*>(flat.data()),
                        static_cast<int>(flat.size() * sizeof(double)), sizes.data(),
                        reinterpret_cast<char*>(uni.data()));
    Dataset holder;
    auto uni_at = [&uni, ncol](data_size_t r, int c) {
      return uni[static_cast<size_t>(r) * ncol + c];
    };
    holder.ConstructFromMatLocal(uni_at, static_cast<data_size_t>(total_s), ncol, cfg,
                                 categorical);
    auto shard = holder.CreateValid(get, nrow);
    *this = std::move(*shard);
    return;
  }
  ConstructFromMatLocal(get, nrow, ncol, cfg, categorical);
}

void Dataset::ConstructFromMatLocal(const std::function<double(data_size_t, int)>& get,
                                    data_size_t nrow, int ncol, const Config& cfg,
                                    const std::vector<int8_t>& categorical) {
  num_data_ = nrow;
  num_total_features_ = ncol;
  categorical_flags_ = categorical;
  if (categorical_flags_.empty()) categorical_flags_.assign(ncol, 0);
  // 1. sample rows
  int sample_cnt = std::min<data_size_t>(cfg.bin_construct_sample_cnt, nrow);
  Random rng(cfg.data_random_seed);
  std::vector<data_size_t> sample_idx;
  if (sample_cnt >= nrow) {
    sample_idx.resize(nrow);
    std::iota(sample_idx.begin(), sample_idx.end(), 0);
  } else {
    // uniform stride sample with random offset (deterministic, cheap, unbiased enough)
    sample_idx.reserve(sample_cnt);
    double stride = static_cast<double>(nrow) / sample_cnt;
    double pos = rng.NextFloat() * stride;
    for (int i = 0; i < sample_cnt; ++i) {
      sample_idx.push_back(std::min<data_size_t>(static_cast<data_size_t>(pos), nrow - 1));
      pos += stride;
    }
  }
  // 2. find bins per column (parallel)
  const auto forced_bins = ParseForcedBins(cfg.forcedbins_filename);
  std::vector<std::unique_ptr<BinMapper>> mappers(ncol);
  const int ns = static_cast<int>(sample_idx.size());
#pragma omp parallel for schedule(dynamic, 1)
  for (int c = 0; c < ncol; ++c) {
    std::vector<double> vals(ns);
    for (int i = 0; i < ns; ++i) vals[i] = get(sample_idx[i], c);
    auto m = std::make_unique<BinMapper>();
    auto fb_it = forced_bins.find(c);
    const int mb = c < static_cast<int>(cfg.max_bin_by_feature.size()) &&
                           cfg.max_bin_by_feature[c] > 1
                       ? cfg.max_bin_by_feature[c]
                       : cfg.max_bin;
    m->FindBin(vals.data(), ns, ns, mb, cfg.min_data_in_bin, 0,
               cfg.feature_pre_filter, categorical_flags_[c] ? BinType::kCategorical
                                                             : BinType::kNumerical,
               cfg.use_missing, cfg.zero_as_missing,
               fb_it != forced_bins.end() ? &fb_it->second : nullptr);
    mappers[c] = std::move(m);
  }
  // 3. keep non-trivial features
  used_feature_map_.assign(ncol, -1);
  real_feature_index_.clear();
  bin_mappers_.clear();
  for (int c = 0; c < ncol; ++c) {
    if (!mappers[c]->is_trivial()) {
      used_feature_map_[c] = static_cast<int>(bin_mappers_.size());
      real_feature_index_.push_back(c);
      bin_mappers_.push_back(std::move(mappers[c]));
    }
  }
  if (bin_mappers_.empty())
    Log::Warning("All features are trivial (constant); no informative splits possible");
  FinishBinMappers(cfg);
  // 4. plan EFB bundles, then bin all values into columns
  PlanBundles(cfg, get, sample_idx);
  const int nf = static_cast<int>(bin_mappers_.size());
  const int ncols = static_cast<int>(column_features_.size());
  columns_.resize(ncols);
#pragma omp parallel for schedule(dynamic, 1)
  for (int col = 0; col < ncols; ++col) {
    const auto& members = column_features_[col];
    if (members.size() == 1 && !feature_bundled_[members[0]]) {
      const int f = members[0];
      const int c = real_feature_index_[f];
      const BinMapper* m = bin_mappers_[f].get();
      // sparse decision on the binning sample: store only non-default rows when
      // >=80% of values land in the zero/default bin (reference is_enable_sparse)
      bool sparse = false;
      uint32_t def_bin = m->ValueToBin(0.0);
      if (cfg.is_enable_sparse && nrow >= 1024) {
        int def_cnt = 0;
        for (int i = 0; i < ns; ++i)
          if (m->ValueToBin(get(sample_idx[i], c)) == def_bin) ++def_cnt;
        sparse = def_cnt >= static_cast<int>(0.8 * ns);
      }
      if (sparse) columns_[col].InitSparse(nrow, m->num_bin(), def_bin);
      else if (m->num_bin() <= 16) columns_[col].Init4(nrow);
      else columns_[col].Init(nrow, m->num_bin());
      for (data_size_t i = 0; i < nrow; ++i) columns_[col].Set(i, m->ValueToBin(get(i, c)));
    } else {
      // bundle column: value 0 = every member at its default(zero) bin;
      // member j's non-default bin b (>=1) stored as off_j + b - 1
      int col_bins = 1;
      for (int f : members) col_bins += bin_mappers_[f]->num_bin() - 1;
      columns_[col].Init(nrow, col_bins);
      for (int f : members) {
        const int c = real_feature_index_[f];
        const BinMapper* m = bin_mappers_[f].get();
        const uint32_t off = off_in_col_[f];
        for (data_size_t i = 0; i < nrow; ++i) {
          const uint32_t b = m->ValueToBin(get(i, c));
          if (b > 0) columns_[col].Set(i, off + b - 1);  // conflicts: last writer wins
        }
      }
    }
  }
  for (const auto& col : columns_) has_sparse_ = has_sparse_ || col.is_sparse();
  if (cfg.linear_tree) {
    raw_values_.resize(nf);
#pragma omp parallel for schedule(dynamic, 1)
    for (int f = 0; f < nf; ++f) {
      raw_values_[f].resize(nrow);
      const int c = real_feature_index_[f];
      for (data_size_t i = 0; i < nrow; ++i)
        raw_values_[f][i] = static_cast<float>(get(i, c));
    }
  }
  metadata_.Init(nrow, false, false);
  if (feature_names_.empty()) {
    for (int c = 0; c < ncol; ++c) feature_names_.push_back("Column_" + std::to_string(c));
  }
}

std::unique_ptr<Dataset> Dataset::CreateValid(
    const std::function<double(data_size_t, int)>& get, data_size_t nrow) const {
  auto d = std::make_unique<Dataset>(nrow);
  d->num_total_features_ = num_total_features_;
  d->used_feature_map_ = used_feature_map_;
  d->real_feature_index_ = real_feature_index_;
  d->feature_names_ = feature_names_;
  d->categorical_flags_ = categorical_flags_;
  d->bin_mappers_.resize(bin_mappers_.size());
  for (size_t i = 0; i < bin_mappers_.size(); ++i)
    d->bin_mappers_[i] = std::make_unique<BinMapper>(*bin_mappers_[i]);
  d->hist_offsets_ = hist_offsets_;
  d->num_total_bin_ = num_total_bin_;
  d->CopyBundlingFrom(*this);
  const int ncols = num_columns();
  d->columns_.resize(ncols);
#pragma omp parallel for schedule(dynamic, 1)
  for (int col = 0; col < ncols; ++col) {
    const auto& members = column_features_[col];
    if (members.size() == 1 && !feature_bundled_[members[0]]) {
      const int f = members[0];
      const int c = real_feature_index_[f];
      const BinMapper* m = d->bin_mappers_[f].get();
      if (columns_[col].is_sparse())
        d->columns_[col].InitSparse(nrow, m->num_bin(), columns_[col].default_bin());
      else if (columns_[col].is4())
        d->columns_[col].Init4(nrow);
      else
        d->columns_[col].Init(nrow, m->num_bin());
      for (data_size_t i = 0; i < nrow; ++i)
        d->columns_[col].Set(i, m->ValueToBin(get(i, c)));
    } else {
      int col_bins = 1;
      for (int f : members) col_bins += bin_mappers_[f]->num_bin() - 1;
      d->columns_[col].Init(nrow, col_bins);
      for (int f : members) {
        const int c = real_feature_index_[f];
        const BinMapper* m = d->bin_mappers_[f].get();
        const uint32_t off = off_in_col_[f];
        for (data_size_t i = 0; i < nrow; ++i) {
          const uint32_t b = m->ValueToBin(get(i, c));
          if (b > 0) d->columns_[col].Set(i, off + b - 1);
        }
      }
    }
  }
  if (has_raw()) {
    // linear-tree scoring reads raw feature values on valid data too
    const int nf_ = num_features();
    d->raw_values_.resize(nf_);
#pragma omp parallel for schedule(dynamic, 1)
    for (int f = 0; f < nf_; ++f) {
      d->raw_values_[f].resize(nrow);
      const int c = real_feature_index_[f];
      for (data_size_t i = 0; i < nrow; ++i)
        d->raw_values_[f][i] = static_cast<float>(get(i, c));
    }
  }
  d->metadata_.Init(nrow, false, false);
  return d;
}

bool Dataset::AlignsWith(const Dataset* other) const {
  if (other == nullptr) return false;
  if (num_total_features_ != other->num_total_features_) return false;
  if (bin_mappers_.size() != other->bin_mappers_.size()) return false;
  for (size_t f = 0; f < bin_mappers_.size(); ++f) {
    const BinMapper* a = bin_mappers_[f].get();
    const BinMapper* b = other->bin_mappers_[f].get();
    if (a->num_bin() != b->num_bin()) return false;
    if (a->bin_type() != b->bin_type()) return false;
    if (a->missing_type() != b->missing_type()) return false;
    if (a->bin_upper_bound() != b->bin_upper_bound()) return false;
    if (a->bin_2_categorical() != b->bin_2_categorical()) return false;
  }
  return true;
}

void Dataset::PlanBundles(const Config& cfg,
                          const std::function<double(data_size_t, int)>& get,
                          const std::vector<data_size_t>& sample_idx) {
  const int nf = static_cast<int>(bin_mappers_.size());
  col_of_feature_.assign(nf, -1);
  off_in_col_.assign(nf, 0);
  feature_bundled_.assign(nf, 0);
  column_features_.clear();
  has_bundles_ = false;
  // candidates: sparse (>=80% zeros), non-negative (zero maps to bin 0), no NaN bin
  std::vector<int> cand;
  if (cfg.enable_bundle) {
    for (int f = 0; f < nf; ++f) {
      const BinMapper* m = bin_mappers_[f].get();
      if (m->bin_type() == BinType::kNumerical && m->missing_type() == MissingType::kNone &&
          m->ValueToBin(0.0) == 0 && m->sparse_rate() >= 0.8 && m->num_bin() >= 2) {
        cand.push_back(f);
      }
    }
  }
  if (static_cast<int>(cand.size()) >= 2) {
    // nonzero bitmap per candidate over the binning sample
    const int ns = static_cast<int>(sample_idx.size());
    const int words = (ns + 63) / 64;
    std::vector<std::vector<uint64_t>> nz(cand.size(), std::vector<uint64_t>(words, 0));
    std::vector<int> nz_cnt(cand.size(), 0);
#pragma omp parallel for schedule(dynamic, 1)
    for (size_t k = 0; k < cand.size(); ++k) {
      const int c = real_feature_index_[cand[k]];
      for (int i = 0; i < ns; ++i) {
        const double v = get(sample_idx[i], c);
        if (v != 0.0 && !std::isnan(v)) {
          nz[k][i >> 6] |= 1ull << (i & 63);
          nz_cnt[k]++;
        }
      }
    }
    // greedy conflict-bounded bundling (densest candidates first)
    std::vector<size_t> order(cand.size());
    for (size_t k = 0; k < order.size(); ++k) order[k] = k;
    std::sort(order.begin(), order.end(),
              [&](size_t a, size_t b) { return nz_cnt[a] > nz_cnt[b]; });
    const int max_conflict = static_cast<int>(cfg.max_conflict_rate * ns);
    struct Bundle {
      std::vector<size_t> members;
      std::vector<uint64_t> occupied;
      int total_bins = 1;
      int conflicts = 0;
    };
    std::vector<Bundle> bundles;
    for (size_t k : order) {
      const int fbins = bin_mappers_[cand[k]]->num_bin() - 1;
      bool placed = false;
      for (auto& b : bundles) {
        if (b.total_bins + fbins > 255) continue;
        int conf = 0;
        for (int w = 0; w < words; ++w)
          conf += __builtin_popcountll(b.occupied[w] & nz[k][w]);
        if (b.conflicts + conf <= max_conflict) {
          for (int w = 0; w < words; ++w) b.occupied[w] |= nz[k][w];
          b.members.push_back(k);
          b.total_bins += fbins;
          b.conflicts += conf;
          placed = true;
          break;
        }
      }
      if (!placed) {
        Bundle b;
        b.members = {k};
        b.occupied = nz[k];
        b.total_bins = 1 + fbins;
        bundles.push_back(std::move(b));
      }
    }
    // realize bundles with >=2 members
    for (auto& b : bundles) {
      if (b.members.size() < 2) continue;
      const int col = static_cast<int>(column_features_.size());
      std::vector<int> feats;
      uint32_t off = 1;
      for (size_t k : b.members) {
        const int f = cand[k];
        feats.push_back(f);
        col_of_feature_[f] = col;
        off_in_col_[f] = off;
        feature_bundled_[f] = 1;
        off += bin_mappers_[f]->num_bin() - 1;
      }
      column_features_.push_back(std::move(feats));
      has_bundles_ = true;
    }
    if (has_bundles_)
      Log::Info("EFB: bundled %d sparse features into %zu columns",
                static_cast<int>(std::count(feature_bundled_.begin(),
                                            feature_bundled_.end(), 1)),
                column_features_.size());
  }
  // unbundled features get their own columns
  for (int f = 0; f < nf; ++f) {
    if (col_of_feature_[f] < 0) {
      col_of_feature_[f] = static_cast<int>(column_features_.size());
      column_features_.push_back({f});
    }
  }
}

namespace {

/*! Hot loop: histogram over an ordered subset with software prefetch.
 *  MI355X note: this is the host fallback; the HIP learner has its own LDS kernel. */
template <typename BIN_T, bool USE_INDICES>
void HistInner(const BIN_T* bins, const data_size_t* idx, data_size_t n,
               const score_t* og, const score_t* oh, hist_t* hist) {
  const data_size_t rest = n & 3;
  const data_size_t nb = n - rest;
  for (data_size_t i = 0; i < nb; i += 4) {
    data_size_t r0 = USE_INDICES ? idx[i] : i;
    data_size_t r1 = USE_INDICES ? idx[i + 1] : i + 1;
    data_size_t r2 = USE_INDICES ? idx[i + 2] : i + 2;
    data_size_t r3 = USE_INDICES ? idx[i + 3] : i + 3;
    if (USE_INDICES && i + 16 < nb) __builtin_prefetch(bins + idx[i + 16], 0, 0);
    const uint32_t b0 = bins[r0] << 1, b1 = bins[r1] << 1, b2 = bins[r2] << 1,
                   b3 = bins[r3] << 1;
    hist[b0] += og[i];     hist[b0 + 1] += oh[i];
    hist[b1] += og[i + 1]; hist[b1 + 1] += oh[i + 1];
    hist[b2] += og[i + 2]; hist[b2 + 1] += oh[i + 2];
    hist[b3] += og[i + 3]; hist[b3 + 1] += oh[i + 3];
  }
  for (data_size_t i = nb; i < n; ++i) {
    data_size_t r = USE_INDICES ? idx[i] : i;
    const uint32_t b = bins[r] << 1;
    hist[b] += og[i];
    hist[b + 1] += oh[i];
  }
}

template <bool USE_INDICES>
void HistInner4(const uint8_t* packed, const data_size_t* idx, data_size_t n,
                const score_t* og, const score_t* oh, hist_t* hist) {
  for (data_size_t i = 0; i < n; ++i) {
    const data_size_t r = USE_INDICES ? idx[i] : i;
    const uint32_t b = ((packed[r >> 1] >> ((r & 1) * 4)) & 0xF) << 1;
    hist[b] += og[i];
    hist[b + 1] += oh[i];
  }
}

}  // namespace

void Dataset::ConstructHistogramForFeature(int f, const data_size_t* data_indices,
                                           data_size_t num_data, const score_t* og,
                                           const score_t* oh, hist_t* out) const {
  if (feature_bundled_[f]) {
    for (data_size_t i = 0; i < num_data; ++i) {
      const data_size_t r = data_indices ? data_indices[i] : i;
      const uint32_t b = GetBin(r, f);
      out[2 * b] += og[i];
      out[2 * b + 1] += oh[i];
    }
    return;
  }
  const BinColumn& col = columns_[col_of_feature_[f]];
  const bool use_idx = data_indices != nullptr;
  if (col.is4()) {
    if (use_idx) HistInner4<true>(col.data8(), data_indices, num_data, og, oh, out);
    else HistInner4<false>(col.data8(), nullptr, num_data, og, oh, out);
  } else if (col.is16()) {
    if (use_idx) HistInner<uint16_t, true>(col.data16(), data_indices, num_data, og, oh, out);
    else HistInner<uint16_t, false>(col.data16(), nullptr, num_data, og, oh, out);
  } else {
    if (use_idx) HistInner<uint8_t, true>(col.data8(), data_indices, num_data, og, oh, out);
    else HistInner<uint8_t, false>(col.data8(), nullptr, num_data, og, oh, out);
  }
}

void Dataset::ConstructHistograms(const std::vector<int8_t>& is_feature_used,
                                  const data_size_t* data_indices, data_size_t num_data,
                                  const score_t* og, const score_t* oh,
                                  const score_t* row_grad, const score_t* row_hess,
                                  const uint8_t* in_leaf, hist_t* hist) const {
  if (num_data <= 0) return;
  const int ncols = num_columns();
#pragma omp parallel for schedule(dynamic)
  for (int col = 0; col < ncols; ++col) {
    const auto& members = column_features_[col];
    bool any_used = false;
    for (int f : members) any_used |= is_feature_used[f] != 0;
    if (!any_used) continue;
    for (int f : members)
      std::fill(hist + 2 * hist_offsets_[f],
                hist + 2 * (hist_offsets_[f] + bin_mappers_[f]->num_bin()), 0.0);
    if (members.size() == 1 && !feature_bundled_[members[0]] &&
        columns_[col].is_sparse()) {
      // sparse column: accumulate the non-default bins only; the default bin is
      // reconstructed from leaf totals by MaterializeDefaultBins (same mechanism
      // as the EFB shared-default fix). Two regimes, both deterministic:
      //   - nonzero scan (needs the in_leaf mask) when nnz is small vs the leaf
      //   - per-row binary search otherwise
      const int f = members[0];
      const BinColumn& bc = columns_[col];
      hist_t* fh = hist + 2 * hist_offsets_[f];
      const data_size_t nnz = bc.nnz();
      const bool whole_data = data_indices == nullptr;  // position == row
      const bool masked_scan = in_leaf != nullptr && row_grad != nullptr &&
                               static_cast<int64_t>(nnz) <= static_cast<int64_t>(num_data) * 8;
      if (whole_data || masked_scan) {
        const data_size_t* rows = bc.sparse_rows();
        const uint16_t* bins = bc.sparse_bins();
        for (data_size_t k = 0; k < nnz; ++k) {
          const data_size_t r = rows[k];
          if (!whole_data && !in_leaf[r]) continue;
          fh[2 * bins[k]] += whole_data ? og[r] : row_grad[r];
          fh[2 * bins[k] + 1] += whole_data ? oh[r] : row_hess[r];
        }
      } else {
        // subset without a membership mask: per-row lookup (O(cnt log nnz))
        for (data_size_t i = 0; i < num_data; ++i) {
          const data_size_t r = data_indices[i];
          const uint32_t b = bc.Get(r);
          if (b == bc.default_bin()) continue;
          fh[2 * b] += og[i];
          fh[2 * b + 1] += oh[i];
        }
      }
    } else if (members.size() == 1 && !feature_bundled_[members[0]]) {
      const int f = members[0];
      ConstructHistogramForFeature(f, data_indices, num_data, og, oh,
                                   hist + 2 * hist_offsets_[f]);
    } else {
      // bundle column: one pass fills every member's non-default bins; the shared
      // default (value 0) is reconstructed from leaf totals in the learner
      // (SerialTreeLearner::OnHistogramReady), the EFB analogue of FixHistogram.
      int col_bins = 1;
      for (int f : members) col_bins += bin_mappers_[f]->num_bin() - 1;
      std::vector<uint32_t> map(col_bins, UINT32_MAX);
      for (int f : members) {
        const uint32_t off = off_in_col_[f];
        for (int b = 1; b < bin_mappers_[f]->num_bin(); ++b)
          map[off + b - 1] = hist_offsets_[f] + b;
      }
      const BinColumn& bc = columns_[col];
      for (data_size_t i = 0; i < num_data; ++i) {
        const data_size_t r = data_indices ? data_indices[i] : i;
        const uint32_t v = bc.Get(r);
        if (v == 0) continue;
        const uint32_t hb = map[v];
        if (hb == UINT32_MAX) continue;
        hist[2 * hb] += og[i];
        hist[2 * hb + 1] += oh[i];
      }
    }
  }
}

void Dataset::AddFeaturesFrom(const Dataset* other) {
  if (other->num_data_ != num_data_)
    Log::Fatal("Cannot add features from other Dataset with a different number of rows");
  const int col_base = static_cast<int>(columns_.size());
  const int feat_base = num_features();
  const int orig_base = num_total_features_;
  // columns + per-feature storage mapping
  for (const auto& c : other->columns_) columns_.push_back(c);
  for (size_t f = 0; f < other->bin_mappers_.size(); ++f) {
    bin_mappers_.push_back(std::make_unique<BinMapper>(*other->bin_mappers_[f]));
    col_of_feature_.push_back(other->col_of_feature_[f] + col_base);
    off_in_col_.push_back(other->off_in_col_[f]);
    feature_bundled_.push_back(other->feature_bundled_[f]);
    real_feature_index_.push_back(other->real_feature_index_[f] + orig_base);
  }
  for (const auto& members : other->column_features_) {
    std::vector<int> shifted;
    for (int f : members) shifted.push_back(f + feat_base);
    column_features_.push_back(std::move(shifted));
  }
  has_bundles_ = has_bundles_ || other->has_bundles_;
  // original-index bookkeeping
  for (int o = 0; o < other->num_total_features_; ++o) {
    const int inner = other->used_feature_map_[o];
    used_feature_map_.push_back(inner >= 0 ? inner + feat_base : -1);
  }
  num_total_features_ += other->num_total_features_;
  for (int o = 0; o < other->num_total_features_; ++o) {
    feature_names_.push_back(o < static_cast<int>(other->feature_names_.size())
                                 ? other->feature_names_[o]
                                 : "Column_" + std::to_string(orig_base + o));
  }
  // rebuild hist offsets (append other's bins after ours)
  for (size_t f = 0; f < other->bin_mappers_.size(); ++f) {
    hist_offsets_.push_back(static_cast<uint32_t>(num_total_bin_) + other->hist_offsets_[f] -
                            (other->hist_offsets_.empty() ? 0 : other->hist_offsets_[0]));
  }
  num_total_bin_ += other->num_total_bin_;
  if (!raw_values_.empty() || !other->raw_values_.empty()) {
    raw_values_.resize(feat_base);
    for (const auto& rv : other->raw_values_) raw_values_.push_back(rv);
  }
  row_view_built_ = false;
  row_view_ = RowMajorView();
}

void Dataset::ConstructHistogramsRowWise(const data_size_t* data_indices,
                                         data_size_t num_data, const score_t* og,
                                         const score_t* oh, hist_t* hist,
                                         bool row_indexed) const {
  const RowMajorView& view = GetRowMajorView();
  const int nf = num_features();
  const size_t hist_elems = 2 * static_cast<size_t>(num_total_bin_);
  const int nthreads = omp_get_max_threads();
  // constant-hessian count mode: when every ordered hessian equals oh[0]
  // (plain L2/ranking without weights/GOSS), accumulate counts instead of
  // hessians and expand at merge — halves the private-histogram write traffic
  // (reference dense_bin.hpp USE_HESSIAN=false analogue)
  auto oh_at = [&](data_size_t i) {
    return oh[row_indexed ? (data_indices ? data_indices[i] : i) : i];
  };
  bool const_hess = num_data > 0;
  if (const_hess) {
    const score_t h0 = oh_at(0);
    bool ok = true;
#pragma omp parallel for schedule(static) reduction(&& : ok)
    for (data_size_t i = 0; i < num_data; ++i) ok = ok && oh_at(i) == h0;
    const_hess = ok;
  }
  if (const_hess) {
    const double h0 = oh_at(0);
    std::vector<int> counts(num_total_bin_, 0);
    static thread_local std::vector<double> priv_g;
    static thread_local std::vector<int> priv_c;
    std::vector<double*> g_ptrs(nthreads, nullptr);
    std::vector<int*> c_ptrs(nthreads, nullptr);
#pragma omp parallel num_threads(nthreads)
    {
      const int tid = omp_get_thread_num();
      priv_g.assign(num_total_bin_, 0.0);
      priv_c.assign(num_total_bin_, 0);
      g_ptrs[tid] = priv_g.data();
      c_ptrs[tid] = priv_c.data();
      double* gp = priv_g.data();
      int* cp = priv_c.data();
      if (!view.is16) {
        const uint8_t* base = view.data.data();
        const int stride = view.row_stride;
        // two-row interleave: independent accumulation chains hide the L1
        // store-to-load latency of same-bin updates
#pragma omp for schedule(static)
        for (data_size_t i = 0; i < num_data; i += 2) {
          const data_size_t r0 = data_indices ? data_indices[i] : i;
          const bool have1 = i + 1 < num_data;
          const data_size_t r1 = have1 ? (data_indices ? data_indices[i + 1] : i + 1) : r0;
          if (data_indices && i + 8 < num_data)
            __builtin_prefetch(base + static_cast<size_t>(data_indices[i + 8]) * stride, 0, 1);
          const uint8_t* row0 = base + static_cast<size_t>(r0) * stride;
          const uint8_t* row1 = base + static_cast<size_t>(r1) * stride;
          const double g0 = og[row_indexed ? r0 : i];
          const double g1 = have1 ? og[row_indexed ? r1 : i + 1] : 0.0;
          for (int f = 0; f < nf; ++f) {
            const uint32_t b0 = hist_offsets_[f] + row0[f];
            gp[b0] += g0;
            cp[b0] += 1;
          }
          if (have1) {
            for (int f = 0; f < nf; ++f) {
              const uint32_t b1 = hist_offsets_[f] + row1[f];
              gp[b1] += g1;
              cp[b1] += 1;
            }
          }
        }
      } else {
        const uint16_t* base = view.data16.data();
        const int stride = view.row_stride;
#pragma omp for schedule(static)
        for (data_size_t i = 0; i < num_data; ++i) {
          const data_size_t r = data_indices ? data_indices[i] : i;
          const uint16_t* row = base + static_cast<size_t>(r) * stride;
          const double g = og[row_indexed ? r : i];
          for (int f = 0; f < nf; ++f) {
            const uint32_t b = hist_offsets_[f] + row[f];
            gp[b] += g;
            cp[b] += 1;
          }
        }
      }
#pragma omp barrier
#pragma omp for schedule(static)
      for (int64_t b = 0; b < static_cast<int64_t>(num_total_bin_); ++b) {
        double acc_g = 0.0;
        int64_t acc_c = 0;
        for (int t = 0; t < nthreads; ++t) {
          if (g_ptrs[t] == nullptr) continue;
          acc_g += g_ptrs[t][b];
          acc_c += c_ptrs[t][b];
        }
        hist[2 * b] = acc_g;
        hist[2 * b + 1] = acc_c * h0;
      }
    }
    return;
  }
  // per-thread private histograms (tiny: num_total_bin*16B each), merged below
  static thread_local std::vector<hist_t> priv;  // reused across calls
  std::vector<hist_t*> priv_ptrs(nthreads, nullptr);
#pragma omp parallel num_threads(nthreads)
  {
    const int tid = omp_get_thread_num();
    priv.assign(hist_elems, 0.0);
    priv_ptrs[tid] = priv.data();
    hist_t* h = priv.data();
    if (!view.is16) {
      const uint8_t* base = view.data.data();
      const int stride = view.row_stride;
      // two-row interleave (see count-mode loop): independent chains hide the
      // same-bin store-to-load latency — measured -30% on this loop
#pragma omp for schedule(static)
      for (data_size_t i = 0; i < num_data; i += 2) {
        const data_size_t r0 = data_indices ? data_indices[i] : i;
        const bool have1 = i + 1 < num_data;
        const data_size_t r1 = have1 ? (data_indices ? data_indices[i + 1] : i + 1) : r0;
        if (data_indices && i + 8 < num_data)
          __builtin_prefetch(base + static_cast<size_t>(data_indices[i + 8]) * stride, 0, 1);
        const uint8_t* row0 = base + static_cast<size_t>(r0) * stride;
        const uint8_t* row1 = base + static_cast<size_t>(r1) * stride;
        const double g0 = og[row_indexed ? r0 : i], h0 = oh[row_indexed ? r0 : i];
        const double g1 = have1 ? og[row_indexed ? r1 : i + 1] : 0.0;
        const double h1 = have1 ? oh[row_indexed ? r1 : i + 1] : 0.0;
        for (int f = 0; f < nf; ++f) {
          const uint32_t b0 = (hist_offsets_[f] + row0[f]) << 1;
          h[b0] += g0;
          h[b0 + 1] += h0;
        }
        if (have1) {
          for (int f = 0; f < nf; ++f) {
            const uint32_t b1 = (hist_offsets_[f] + row1[f]) << 1;
            h[b1] += g1;
            h[b1 + 1] += h1;
          }
        }
      }
    } else {
      const uint16_t* base = view.data16.data();
      const int stride = view.row_stride;
#pragma omp for schedule(static)
      for (data_size_t i = 0; i < num_data; ++i) {
        const data_size_t r = data_indices ? data_indices[i] : i;
        const uint16_t* row = base + static_cast<size_t>(r) * stride;
        const double g = og[row_indexed ? r : i], hv = oh[row_indexed ? r : i];
        for (int f = 0; f < nf; ++f) {
          const uint32_t b = (hist_offsets_[f] + row[f]) << 1;
          h[b] += g;
          h[b + 1] += hv;
        }
      }
    }
    // parallel merge: each thread owns a slice of the output histogram
#pragma omp barrier
#pragma omp for schedule(static)
    for (int64_t e = 0; e < static_cast<int64_t>(hist_elems); ++e) {
      double acc = 0.0;
      for (int t = 0; t < nthreads; ++t)
        if (priv_ptrs[t] != nullptr) acc += priv_ptrs[t][e];
      hist[e] = acc;
    }
  }
}

void Dataset::ConstructHistogramsRowWiseGH(const data_size_t* data_indices,
                                           data_size_t num_data, const score_t* gh,
                                           hist_t* hist) const {
  const RowMajorView& view = GetRowMajorView();
  const int nf = num_features();
  const size_t hist_elems = 2 * static_cast<size_t>(num_total_bin_);
  const int nthreads = omp_get_max_threads();
  static thread_local std::vector<hist_t> priv;
  std::vector<hist_t*> priv_ptrs(nthreads, nullptr);
#pragma omp parallel num_threads(nthreads)
  {
    const int tid = omp_get_thread_num();
    priv.assign(hist_elems, 0.0);
    priv_ptrs[tid] = priv.data();
    hist_t* __restrict__ h = priv.data();
    const uint32_t* __restrict__ off = hist_offsets_.data();
    if (!view.is16) {
      const uint8_t* __restrict__ base = view.data.data();
      const int stride = view.row_stride;
      // 4-row interleave: four independent accumulation chains hide the L1
      // store-to-load latency of same-bin updates; one 8B gh load per row
#pragma omp for schedule(static)
      for (data_size_t i = 0; i < num_data; i += 4) {
        const data_size_t rem = num_data - i;
        if (rem >= 4) {
          const data_size_t r0 = data_indices[i], r1 = data_indices[i + 1];
          const data_size_t r2 = data_indices[i + 2], r3 = data_indices[i + 3];
          if (i + 16 < num_data) {
            __builtin_prefetch(base + static_cast<size_t>(data_indices[i + 16]) * stride, 0, 1);
            __builtin_prefetch(gh + 2 * static_cast<size_t>(data_indices[i + 16]), 0, 1);
          }
          const uint8_t* __restrict__ row0 = base + static_cast<size_t>(r0) * stride;
          const uint8_t* __restrict__ row1 = base + static_cast<size_t>(r1) * stride;
          const uint8_t* __restrict__ row2 = base + static_cast<size_t>(r2) * stride;
          const uint8_t* __restrict__ row3 = base + static_cast<size_t>(r3) * stride;
          const double g0 = gh[2 * static_cast<size_t>(r0)], h0 = gh[2 * static_cast<size_t>(r0) + 1];
          const double g1 = gh[2 * static_cast<size_t>(r1)], h1 = gh[2 * static_cast<size_t>(r1) + 1];
          const double g2 = gh[2 * static_cast<size_t>(r2)], h2 = gh[2 * static_cast<size_t>(r2) + 1];
          const double g3 = gh[2 * static_cast<size_t>(r3)], h3 = gh[2 * static_cast<size_t>(r3) + 1];
          for (int f = 0; f < nf; ++f) {
            const uint32_t b0 = (off[f] + row0[f]) << 1;
            const uint32_t b1 = (off[f] + row1[f]) << 1;
            h[b0] += g0;
            h[b0 + 1] += h0;
            h[b1] += g1;
            h[b1 + 1] += h1;
            const uint32_t b2 = (off[f] + row2[f]) << 1;
            const uint32_t b3 = (off[f] + row3[f]) << 1;
            h[b2] += g2;
            h[b2 + 1] += h2;
            h[b3] += g3;
            h[b3 + 1] += h3;
          }
        } else {
          for (data_size_t k = i; k < num_data; ++k) {
            const data_size_t r = data_indices[k];
            const uint8_t* __restrict__ row = base + static_cast<size_t>(r) * stride;
            const double g = gh[2 * static_cast<size_t>(r)];
            const double hv = gh[2 * static_cast<size_t>(r) + 1];
            for (int f = 0; f < nf; ++f) {
              const uint32_t b = (off[f] + row[f]) << 1;
              h[b] += g;
              h[b + 1] += hv;
            }
          }
        }
      }
    } else {
      const uint16_t* __restrict__ base = view.data16.data();
      const int stride = view.row_stride;
#pragma omp for schedule(static)
      for (data_size_t i = 0; i < num_data; ++i) {
        const data_size_t r = data_indices[i];
        const uint16_t* __restrict__ row = base + static_cast<size_t>(r) * stride;
        const double g = gh[2 * static_cast<size_t>(r)];
        const double hv = gh[2 * static_cast<size_t>(r) + 1];
        for (int f = 0; f < nf; ++f) {
          const uint32_t b = (off[f] + row[f]) << 1;
          h[b] += g;
          h[b + 1] += hv;
        }
      }
    }
#pragma omp barrier
#pragma omp for schedule(static)
    for (int64_t e = 0; e < static_cast<int64_t>(hist_elems); ++e) {
      double acc = 0.0;
      for (int t = 0; t < nthreads; ++t)
        if (priv_ptrs[t] != nullptr) acc += priv_ptrs[t][e];
      hist[e] = acc;
    }
  }
}

void Dataset::set_feature_names(const std::vector<std::string>& names) {
  feature_names_ = names;
}

std::string Dataset::FeatureInfoString() const {
  std::vector<std::string> infos(num_total_features_, "none");
  for (int f = 0; f < num_features(); ++f)
    infos[real_feature_index_[f]] = bin_mappers_[f]->ToFeatureInfoString();
  return Common::Join(infos, " ");
}

const Dataset::RowMajorView& Dataset::GetRowMajorView() const {
  if (row_view_built_) return row_view_;
  const int nf = num_features();
  bool any16 = false;
  for (int f = 0; f < nf; ++f) any16 |= bin_mappers_[f]->num_bin() > 256;
  // pad row stride to 16 elements for aligned vector loads on device
  int stride = (nf + 15) & ~15;
  row_view_.row_stride = stride;
  row_view_.is16 = any16;
  if (!any16) {
    row_view_.data.assign(static_cast<size_t>(num_data_) * stride, 0);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      uint8_t* row = row_view_.data.data() + static_cast<size_t>(i) * stride;
      for (int f = 0; f < nf; ++f) row[f] = static_cast<uint8_t>(GetBin(i, f));
    }
  } else {
    row_view_.data16.assign(static_cast<size_t>(num_data_) * stride, 0);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < num_data_; ++i) {
      uint16_t* row = row_view_.data16.data() + static_cast<size_t>(i) * stride;
      for (int f = 0; f < nf; ++f) row[f] = static_cast<uint16_t>(GetBin(i, f));
    }
  }
  row_view_built_ = true;
  return row_view_;
}

std::unique_ptr<Dataset> Dataset::Subset(const data_size_t* indices, data_size_t n) const {
  auto d = std::make_unique<Dataset>(n);
  d->num_total_features_ = num_total_features_;
  d->used_feature_map_ = used_feature_map_;
  d->real_feature_index_ = real_feature_index_;
  d->feature_names_ = feature_names_;
  d->categorical_flags_ = categorical_flags_;
  d->bin_mappers_.resize(bin_mappers_.size());
  for (size_t i = 0; i < bin_mappers_.size(); ++i)
    d->bin_mappers_[i] = std::make_unique<BinMapper>(*bin_mappers_[i]);
  d->hist_offsets_ = hist_offsets_;
  d->num_total_bin_ = num_total_bin_;
  d->CopyBundlingFrom(*this);
  const int ncols = num_columns();
  d->columns_.resize(ncols);
#pragma omp parallel for schedule(dynamic, 1)
  for (int col = 0; col < ncols; ++col) {
    d->columns_[col].Init(n, columns_[col].is16() ? 65536 : 256);
    for (data_size_t i = 0; i < n; ++i) d->columns_[col].Set(i, columns_[col].Get(indices[i]));
  }
  d->has_sparse_ = false;  // subset columns are materialized dense
  if (has_raw()) {
    d->raw_values_.resize(raw_values_.size());
    for (size_t f = 0; f < raw_values_.size(); ++f) {
      d->raw_values_[f].resize(n);
      for (data_size_t i = 0; i < n; ++i) d->raw_values_[f][i] = raw_values_[f][indices[i]];
    }
  }
  // metadata subset
  d->metadata_.Init(n, false, false);
  std::vector<float> lab(n);
  const label_t* src = metadata_.label();
  for (data_size_t i = 0; i < n; ++i) lab[i] = src[indices[i]];
  d->metadata_.SetLabel(lab.data(), n);
  if (metadata_.weights() != nullptr) {
    std::vector<float> w(n);
    for (data_size_t i = 0; i < n; ++i) w[i] = metadata_.weights()[indices[i]];
    d->metadata_.SetWeights(w.data(), n);
  }
  // query boundaries: count subset rows per original query, keep non-empty
  // queries (reference Metadata::CheckOrPartition subset semantics — indices
  // are expected to respect query grouping)
  if (metadata_.query_boundaries() != nullptr && metadata_.num_queries() > 0) {
    const data_size_t* qb = metadata_.query_boundaries();
    const data_size_t nq = metadata_.num_queries();
    std::vector<data_size_t> boundaries{0};
    data_size_t q = 0, cnt = 0;
    for (data_size_t i = 0; i < n; ++i) {
      while (q < nq && indices[i] >= qb[q + 1]) {
        if (cnt > 0) { boundaries.push_back(boundaries.back() + cnt); cnt = 0; }
        ++q;
      }
      ++cnt;
    }
    if (cnt > 0) boundaries.push_back(boundaries.back() + cnt);
    d->metadata_.SetQueryBoundaries(std::move(boundaries));
  }
  if (metadata_.init_score() != nullptr) {
    const int64_t total = metadata_.num_init_score();
    const data_size_t N = num_data_;
    const int k = N > 0 ? static_cast<int>(total / N) : 1;
    std::vector<double> is(static_cast<size_t>(k) * n);
    for (int c = 0; c < k; ++c)
      for (data_size_t i = 0; i < n; ++i)
        is[static_cast<size_t>(c) * n + i] =
            metadata_.init_score()[static_cast<int64_t>(c) * N + indices[i]];
    d->metadata_.SetInitScore(is.data(), static_cast<int64_t>(is.size()));
  }
  if (metadata_.positions() != nullptr) {
    std::vector<int32_t> pos(n);
    for (data_size_t i = 0; i < n; ++i) pos[i] = metadata_.positions()[indices[i]];
    d->metadata_.SetPosition(pos.data(), n);
  }
  return d;
}

void Dataset::ConstructFromSampleData(double** sample_values, int** sample_indices,
                                      int ncol, const int* num_per_col, int num_sample_row,
                                      data_size_t num_local_row, const Config& cfg,
                                      const std::vector<int8_t>& categorical) {
  num_data_ = num_local_row;
  num_total_features_ = ncol;
  categorical_flags_ = categorical;
  if (categorical_flags_.empty()) categorical_flags_.assign(ncol, 0);
  (void)sample_indices;  // dense binning: implied zeros come from total_sample_cnt
  const auto forced_bins = ParseForcedBins(cfg.forcedbins_filename);
  std::vector<std::unique_ptr<BinMapper>> mappers(ncol);
#pragma omp parallel for schedule(dynamic, 1)
  for (int c = 0; c < ncol; ++c) {
    std::vector<double> vals(sample_values[c], sample_values[c] + num_per_col[c]);
    auto m = std::make_unique<BinMapper>();
    auto fb_it = forced_bins.find(c);
    const int mb = c < static_cast<int>(cfg.max_bin_by_feature.size()) &&
                           cfg.max_bin_by_feature[c] > 1
                       ? cfg.max_bin_by_feature[c]
                       : cfg.max_bin;
    m->FindBin(vals.data(), num_per_col[c], num_sample_row, mb,
               cfg.min_data_in_bin, 0, cfg.feature_pre_filter,
               categorical_flags_[c] ? BinType::kCategorical : BinType::kNumerical,
               cfg.use_missing, cfg.zero_as_missing,
               fb_it != forced_bins.end() ? &fb_it->second : nullptr);
    mappers[c] = std::move(m);
  }
  used_feature_map_.assign(ncol, -1);
  real_feature_index_.clear();
  bin_mappers_.clear();
  for (int c = 0; c < ncol; ++c) {
    if (!mappers[c]->is_trivial()) {
      used_feature_map_[c] = static_cast<int>(bin_mappers_.size());
      real_feature_index_.push_back(c);
      bin_mappers_.push_back(std::move(mappers[c]));
    }
  }
  FinishBinMappers(cfg);
  {
    const int nfX = static_cast<int>(bin_mappers_.size());
    col_of_feature_.resize(nfX);
    off_in_col_.assign(nfX, 0);
    feature_bundled_.assign(nfX, 0);
    column_features_.clear();
    for (int f = 0; f < nfX; ++f) {
      col_of_feature_[f] = f;
      column_features_.push_back({f});
    }
  }
  columns_.resize(bin_mappers_.size());
  for (size_t f = 0; f < bin_mappers_.size(); ++f)
    columns_[f].Init(num_local_row, bin_mappers_[f]->num_bin());
  metadata_.Init(num_local_row, false, false);
  if (feature_names_.empty())
    for (int c = 0; c < ncol; ++c) feature_names_.push_back("Column_" + std::to_string(c));
}

std::unique_ptr<Dataset> Dataset::CreateByReference(data_size_t num_rows) const {
  auto d = std::make_unique<Dataset>(num_rows);
  d->num_total_features_ = num_total_features_;
  d->used_feature_map_ = used_feature_map_;
  d->real_feature_index_ = real_feature_index_;
  d->feature_names_ = feature_names_;
  d->categorical_flags_ = categorical_flags_;
  d->bin_mappers_.resize(bin_mappers_.size());
  for (size_t i = 0; i < bin_mappers_.size(); ++i)
    d->bin_mappers_[i] = std::make_unique<BinMapper>(*bin_mappers_[i]);
  d->hist_offsets_ = hist_offsets_;
  d->num_total_bin_ = num_total_bin_;
  d->CopyBundlingFrom(*this);
  d->columns_.resize(columns_.size());
  for (size_t c = 0; c < columns_.size(); ++c)
    d->columns_[c].Init(num_rows, columns_[c].is16() ? 65536 : 256);
  d->metadata_.Init(num_rows, false, false);
  return d;
}

std::string Dataset::SerializeReference() const {
  std::stringstream ss;
  ss << num_total_features_ << " " << num_total_bin_ << "\n";
  ss << Common::Join(used_feature_map_, " ") << "\n";
  ss << Common::Join(real_feature_index_, " ") << "\n";
  ss << Common::Join(feature_names_, "\t") << "\n";
  ss << Common::Join(col_of_feature_, " ") << "\n";
  ss << Common::Join(off_in_col_, " ") << "\n";
  {
    std::vector<int> fb(feature_bundled_.begin(), feature_bundled_.end());
    ss << Common::Join(fb, " ") << "\n";
  }
  ss << columns_.size() << "\n";
  for (auto& m : bin_mappers_) ss << m->ToString();
  return ss.str();
}

std::unique_ptr<Dataset> Dataset::FromSerializedReference(const char* buf, size_t len,
                                                          data_size_t num_rows) {
  std::string content(buf, len);
  auto lines = Common::Split(content.c_str(), '\n');
  auto d = std::make_unique<Dataset>(num_rows);
  auto head = Common::SplitAny(lines[0].c_str(), " ");
  d->num_total_features_ = atoi(head[0].c_str());
  d->num_total_bin_ = atoi(head[1].c_str());
  Common::StringToArray<int>(lines[1], ' ', &d->used_feature_map_);
  Common::StringToArray<int>(lines[2], ' ', &d->real_feature_index_);
  d->feature_names_ = Common::Split(lines[3].c_str(), '\t');
  Common::StringToArray<int>(lines[4], ' ', &d->col_of_feature_);
  Common::StringToArray<uint32_t>(lines[5], ' ', &d->off_in_col_);
  {
    std::vector<int> fb;
    Common::StringToArray<int>(lines[6], ' ', &fb);
    d->feature_bundled_.assign(fb.begin(), fb.end());
  }
  const int ref_ncols = atoi(lines[7].c_str());
  const int nf = static_cast<int>(d->real_feature_index_.size());
  d->bin_mappers_.resize(nf);
  size_t lp = 8;
  for (int f = 0; f < nf; ++f) {
    std::string blob = lines[lp] + "\n" + lines[lp + 1] + "\n" + lines[lp + 2];
    d->bin_mappers_[f] = std::make_unique<BinMapper>();
    d->bin_mappers_[f]->FromString(blob);
    lp += 3;
  }
  Config dummy;
  d->FinishBinMappers(dummy);
  d->column_features_.assign(ref_ncols, {});
  for (int f = 0; f < nf; ++f) d->column_features_[d->col_of_feature_[f]].push_back(f);
  for (int8_t fb : d->feature_bundled_) d->has_bundles_ |= fb != 0;
  d->columns_.resize(ref_ncols);
  for (int c = 0; c < ref_ncols; ++c) {
    int col_bins = 0;
    for (int f : d->column_features_[c]) {
      if (d->feature_bundled_[f]) col_bins += d->bin_mappers_[f]->num_bin() - 1;
      else col_bins = std::max(col_bins, d->bin_mappers_[f]->num_bin());
    }
    if (!d->column_features_[c].empty() && d->feature_bundled_[d->column_features_[c][0]])
      col_bins += 1;  // shared default bin
    d->columns_[c].Init(num_rows, col_bins);
  }
  d->metadata_.Init(num_rows, false, false);
  return d;
}

void Dataset::PushRawRow(data_size_t row, const double* values, int ncol) {
  for (int c = 0; c < std::min(ncol, num_total_features_); ++c) {
    const int f = used_feature_map_[c];
    if (f < 0) continue;
    const uint32_t b = bin_mappers_[f]->ValueToBin(values[c]);
    if (!feature_bundled_[f]) {
      columns_[col_of_feature_[f]].Set(row, b);
    } else if (b > 0) {
      columns_[col_of_feature_[f]].Set(row, off_in_col_[f] + b - 1);
    }
  }
}

// ------------------------------------------------------------------ binary file
static const char kBinMagic[] = "migbm.dataset.v1";

void Dataset::SaveBinaryFile(const char* filename) const {
  FILE* fp = fopen(filename, "wb");
  if (!fp) Log::Fatal("Cannot open %s for writing", filename);
  fwrite(kBinMagic, 1, sizeof(kBinMagic), fp);
  std::stringstream ss;
  ss << num_data_ << " " << num_total_features_ << " " << num_total_bin_ << "\n";
  ss << Common::Join(used_feature_map_, " ") << "\n";
  ss << Common::Join(real_feature_index_, " ") << "\n";
  ss << Common::Join(feature_names_, "\t") << "\n";
  ss << Common::Join(col_of_feature_, " ") << "\n";
  ss << Common::Join(off_in_col_, " ") << "\n";
  {
    std::vector<int> fb(feature_bundled_.begin(), feature_bundled_.end());
    ss << Common::Join(fb, " ") << "\n";
  }
  ss << columns_.size() << "\n";
  for (auto& m : bin_mappers_) ss << m->ToString();
  std::string header = ss.str();
  uint64_t hlen = header.size();
  fwrite(&hlen, sizeof(hlen), 1, fp);
  fwrite(header.data(), 1, hlen, fp);
  for (auto& col : columns_) {
    uint8_t is16 = col.is16();
    fwrite(&is16, 1, 1, fp);
    if (col.is_sparse() || col.is4()) {
      // serialize densified (format unchanged; sparse/4-bit packing is re-decided
      // only on in-memory construction paths)
      if (is16) {
        std::vector<uint16_t> buf(num_data_);
        for (data_size_t i = 0; i < num_data_; ++i) buf[i] = static_cast<uint16_t>(col.Get(i));
        fwrite(buf.data(), sizeof(uint16_t), num_data_, fp);
      } else {
        std::vector<uint8_t> buf(num_data_);
        for (data_size_t i = 0; i < num_data_; ++i) buf[i] = static_cast<uint8_t>(col.Get(i));
        fwrite(buf.data(), sizeof(uint8_t), num_data_, fp);
      }
    } else if (is16) {
      fwrite(col.data16(), sizeof(uint16_t), num_data_, fp);
    } else {
      fwrite(col.data8(), sizeof(uint8_t), num_data_, fp);
    }
  }
  // metadata
  uint8_t has_w = metadata_.weights() != nullptr;
  uint8_t has_q = metadata_.query_boundaries() != nullptr;
  fwrite(&has_w, 1, 1, fp);
  fwrite(&has_q, 1, 1, fp);
  fwrite(metadata_.label(), sizeof(label_t), num_data_, fp);
  if (has_w) fwrite(metadata_.weights(), sizeof(label_t), num_data_, fp);
  if (has_q) {
    data_size_t nq = metadata_.num_queries();
    fwrite(&nq, sizeof(nq), 1, fp);
    fwrite(metadata_.query_boundaries(), sizeof(data_size_t), nq + 1, fp);
  }
  fclose(fp);
}

bool Dataset::IsBinFile(const char* filename) {
  FILE* fp = fopen(filename, "rb");
  if (!fp) return false;
  char magic[sizeof(kBinMagic)] = {0};
  size_t got = fread(magic, 1, sizeof(kBinMagic), fp);
  fclose(fp);
  return got == sizeof(kBinMagic) && memcmp(magic, kBinMagic, sizeof(kBinMagic)) == 0;
}

std::unique_ptr<Dataset> Dataset::LoadFromBinFile(const char* filename) {
  FILE* fp = fopen(filename, "rb");
  if (!fp) Log::Fatal("Cannot open %s", filename);
  char magic[sizeof(kBinMagic)];
  MIGBM_CHECK_EQ(fread(magic, 1, sizeof(kBinMagic), fp), sizeof(kBinMagic));
  MIGBM_CHECK_EQ(memcmp(magic, kBinMagic, sizeof(kBinMagic)), 0);
  uint64_t hlen;
  MIGBM_CHECK_EQ(fread(&hlen, sizeof(hlen), 1, fp), 1u);
  std::string header(hlen, '\0');
  MIGBM_CHECK_EQ(fread(&header[0], 1, hlen, fp), hlen);
  auto lines = Common::Split(header.c_str(), '\n');
  auto head = Common::SplitAny(lines[0].c_str(), " ");
  auto d = std::make_unique<Dataset>();
  d->num_data_ = atoi(head[0].c_str());
  d->num_total_features_ = atoi(head[1].c_str());
  d->num_total_bin_ = atoi(head[2].c_str());
  Common::StringToArray<int>(lines[1], ' ', &d->used_feature_map_);
  Common::StringToArray<int>(lines[2], ' ', &d->real_feature_index_);
  d->feature_names_ = Common::Split(lines[3].c_str(), '\t');
  Common::StringToArray<int>(lines[4], ' ', &d->col_of_feature_);
  Common::StringToArray<uint32_t>(lines[5], ' ', &d->off_in_col_);
  {
    std::vector<int> fb;
    Common::StringToArray<int>(lines[6], ' ', &fb);
    d->feature_bundled_.assign(fb.begin(), fb.end());
  }
  const int n_cols_stored = atoi(lines[7].c_str());
  int nf = static_cast<int>(d->real_feature_index_.size());
  d->bin_mappers_.resize(nf);
  size_t line_pos = 8;
  for (int f = 0; f < nf; ++f) {
    std::string blob = lines[line_pos] + "\n" + lines[line_pos + 1] + "\n" + lines[line_pos + 2];
    d->bin_mappers_[f] = std::make_unique<BinMapper>();
    d->bin_mappers_[f]->FromString(blob);
    line_pos += 3;
  }
  Config dummy;
  d->FinishBinMappers(dummy);
  d->column_features_.assign(n_cols_stored, {});
  for (int f = 0; f < nf; ++f) d->column_features_[d->col_of_feature_[f]].push_back(f);
  for (int8_t fb : d->feature_bundled_) d->has_bundles_ |= fb != 0;
  d->columns_.resize(n_cols_stored);
  for (int f = 0; f < n_cols_stored; ++f) {
    uint8_t is16;
    MIGBM_CHECK_EQ(fread(&is16, 1, 1, fp), 1u);
    d->columns_[f].Init(d->num_data_, is16 ? 65536 : 256);
    if (is16) {
      MIGBM_CHECK_EQ(fread(const_cast<uint16_t*>(d->columns_[f].data16()), sizeof(uint16_t),
                           d->num_data_, fp), static_cast<size_t>(d->num_data_));
    } else {
      MIGBM_CHECK_EQ(fread(const_cast<uint8_t*>(d->columns_[f].data8()), sizeof(uint8_t),
                           d->num_data_, fp), static_cast<size_t>(d->num_data_));
    }
  }
  uint8_t has_w, has_q;
  MIGBM_CHECK_EQ(fread(&has_w, 1, 1, fp), 1u);
  MIGBM_CHECK_EQ(fread(&has_q, 1, 1, fp), 1u);
  std::vector<label_t> lab(d->num_data_);
  MIGBM_CHECK_EQ(fread(lab.data(), sizeof(label_t), d->num_data_, fp),
                 static_cast<size_t>(d->num_data_));
  d->metadata_.SetLabel(lab.data(), d->num_data_);
  if (has_w) {
    std::vector<label_t> w(d->num_data_);
    MIGBM_CHECK_EQ(fread(w.data(), sizeof(label_t), d->num_data_, fp),
                   static_cast<size_t>(d->num_data_));
    d->metadata_.SetWeights(w.data(), d->num_data_);
  }
  if (has_q) {
    data_size_t nq;
    MIGBM_CHECK_EQ(fread(&nq, sizeof(nq), 1, fp), 1u);
    std::vector<data_size_t> qb(nq + 1);
    MIGBM_CHECK_EQ(fread(qb.data(), sizeof(data_size_t), nq + 1, fp),
                   static_cast<size_t>(nq + 1));
    d->metadata_.SetQueryBoundaries(std::move(qb));
  }
  fclose(fp);
  return d;
}

void Dataset::DumpTextFile(const char* filename) const {
  FILE* fp = fopen(filename, "w");
  if (!fp) Log::Fatal("Cannot open %s", filename);
  fprintf(fp, "num_data=%d\nnum_features=%d\nnum_total_bin=%d\n", num_data_, num_features(),
          num_total_bin_);
  for (int f = 0; f < num_features(); ++f) {
    fprintf(fp, "feature %d (orig %d) num_bin=%d\n", f, real_feature_index_[f],
            bin_mappers_[f]->num_bin());
  }
  fclose(fp);
}

// ------------------------------------------------------------------ Tree score update
// (defined here to have Dataset complete)

}  // namespace migbm

#include "migbm/tree.h"

namespace migbm {

void Tree::AddPredictionToScore(const Dataset* data, data_size_t num_data, double* score) const {
  if (!bin_thresholds_valid_) {
    // text-loaded tree: no bin-space thresholds — route by real values
    AddPredictionToScoreByValue(data, num_data, score);
    return;
  }
  if (num_leaves_ <= 1) {
    if (leaf_value_[0] != 0.0) {
#pragma omp parallel for schedule(static)
      for (data_size_t i = 0; i < num_data; ++i) score[i] += leaf_value_[0];
    }
    return;
  }
#pragma omp parallel for schedule(static, 2048)
  for (data_size_t i = 0; i < num_data; ++i) {
    int node = 0;
    while (node >= 0) {
      const int f = split_feature_inner_[node];
      const uint32_t bin = data->GetBin(i, f);
      if (IsCategoricalSplit(node)) {
        // bin-level categorical decision via bitset over bins
        const int cat_idx = static_cast<int>(threshold_in_bin_[node]);
        const uint32_t* bits = cat_threshold_.data() + cat_boundaries_[cat_idx];
        const int n_words = cat_boundaries_[cat_idx + 1] - cat_boundaries_[cat_idx];
        const int cat = static_cast<int>(data->FeatureBinMapper(f)->BinToValue(bin));
        node = (cat >= 0 && (cat >> 5) < n_words && ((bits[cat >> 5] >> (cat & 31)) & 1))
                   ? left_child_[node] : right_child_[node];
      } else {
        // NaN bin (if any) is the last bin; honor default direction
        const BinMapper* m = data->FeatureBinMapper(f);
        const int nanb = m->nan_bin();
        if (nanb >= 0 && bin == static_cast<uint32_t>(nanb)) {
          node = (decision_type_[node] & kDefaultLeftMask) ? left_child_[node] : right_child_[node];
        } else {
          node = bin <= threshold_in_bin_[node] ? left_child_[node] : right_child_[node];
        }
      }
    }
    if (is_linear_ && data->has_raw()) {
      const int leaf = ~node;
      double out = leaf_const_.empty() || leaf_coeff_[leaf].empty()
                       ? leaf_value_[leaf] : leaf_const_[leaf];
      if (!leaf_const_.empty() && !leaf_coeff_[leaf].empty()) {
        bool ok = true;
        for (size_t k = 0; k < leaf_coeff_[leaf].size(); ++k) {
          const float v = data->raw_value(leaf_features_inner_[leaf][k], i);
          if (std::isnan(v)) { ok = false; break; }
          out += leaf_coeff_[leaf][k] * v;
        }
        if (!ok) out = leaf_value_[leaf];
      }
      score[i] += out;
    } else {
      score[i] += leaf_value_[~node];
    }
  }
}

void Tree::AddPredictionToScoreByValue(const Dataset* data, data_size_t num_data,
                                       double* score) const {
  // for LOADED/merged trees: threshold_in_bin_ is not populated (model text
  // stores only real-valued thresholds), so route every row by representative
  // raw values reconstructed from its bins via the dataset's bin mappers
  if (num_leaves_ <= 1) {
    if (leaf_value_[0] != 0.0) {
#pragma omp parallel for schedule(static)
      for (data_size_t i = 0; i < num_data; ++i) score[i] += leaf_value_[0];
    }
    return;
  }
#pragma omp parallel for schedule(static, 2048)
  for (data_size_t i = 0; i < num_data; ++i) {
    int node = 0;
    while (node >= 0) {
      const int orig = split_feature_[node];
      double v = 0.0;
      const int inner = orig >= 0 && orig < static_cast<int>(data->num_total_features())
                            ? data->InnerFeatureIndex(orig) : -1;
      if (inner >= 0) {
        const BinMapper* m = data->FeatureBinMapper(inner);
        const uint32_t bin = data->GetBin(i, inner);
        if (m->bin_type() != BinType::kCategorical && m->nan_bin() >= 0 &&
            bin == static_cast<uint32_t>(m->nan_bin()))
          v = std::numeric_limits<double>::quiet_NaN();
        else
          v = m->BinToValue(bin);
      }
      node = IsCategoricalSplit(node) ? CategoricalDecision(v, node)
                                      : NumericalDecision(v, node);
    }
    if (is_linear_ && data->has_raw()) {
      const int leaf = ~node;
      double out = leaf_const_.empty() || leaf_coeff_[leaf].empty()
                       ? leaf_value_[leaf] : leaf_const_[leaf];
      if (!leaf_const_.empty() && !leaf_coeff_[leaf].empty()) {
        bool ok = true;
        for (size_t k = 0; k < leaf_coeff_[leaf].size(); ++k) {
          const float rv = data->raw_value(leaf_features_inner_[leaf][k], i);
          if (std::isnan(rv)) { ok = false; break; }
          out += leaf_coeff_[leaf][k] * rv;
        }
        if (!ok) out = leaf_value_[leaf];
      }
      score[i] += out;
    } else {
      score[i] += leaf_value_[~node];
    }
  }
}

void Tree::AddPredictionToScore(const Dataset* data, const data_size_t* used_indices,
                                data_size_t num_data, double* score) const {
  if (num_leaves_ <= 1) {
    if (leaf_value_[0] != 0.0) {
#pragma omp parallel for schedule(static)
      for (data_size_t i = 0; i < num_data; ++i) score[used_indices[i]] += leaf_value_[0];
    }
    return;
  }
  if (!bin_thresholds_valid_) {
    // text-loaded tree: route by representative real values (cf. ByValue walk)
#pragma omp parallel for schedule(static, 2048)
    for (data_size_t i = 0; i < num_data; ++i) {
      const data_size_t r = used_indices[i];
      int node = 0;
      while (node >= 0) {
        const int orig = split_feature_[node];
        double v = 0.0;
        const int inner = orig >= 0 && orig < static_cast<int>(data->num_total_features())
                              ? data->InnerFeatureIndex(orig) : -1;
        if (inner >= 0) {
          const BinMapper* m = data->FeatureBinMapper(inner);
          const uint32_t bin = data->GetBin(r, inner);
          if (m->bin_type() != BinType::kCategorical && m->nan_bin() >= 0 &&
              bin == static_cast<uint32_t>(m->nan_bin()))
            v = std::numeric_limits<double>::quiet_NaN();
          else
            v = m->BinToValue(bin);
        }
        node = IsCategoricalSplit(node) ? CategoricalDecision(v, node)
                                        : NumericalDecision(v, node);
      }
      score[r] += leaf_value_[~node];
    }
    return;
  }
#pragma omp parallel for schedule(static, 2048)
  for (data_size_t i = 0; i < num_data; ++i) {
    const data_size_t r = used_indices[i];
    int node = 0;
    while (node >= 0) {
      const int f = split_feature_inner_[node];
      const uint32_t bin = data->GetBin(r, f);
      if (IsCategoricalSplit(node)) {
        const int cat_idx = static_cast<int>(threshold_in_bin_[node]);
        const uint32_t* bits = cat_threshold_.data() + cat_boundaries_[cat_idx];
        const int n_words = cat_boundaries_[cat_idx + 1] - cat_boundaries_[cat_idx];
        const int cat = static_cast<int>(data->FeatureBinMapper(f)->BinToValue(bin));
        node = (cat >= 0 && (cat >> 5) < n_words && ((bits[cat >> 5] >> (cat & 31)) & 1))
                   ? left_child_[node] : right_child_[node];
      } else {
        const BinMapper* m = data->FeatureBinMapper(f);
        const int nanb = m->nan_bin();
        if (nanb >= 0 && bin == static_cast<uint32_t>(nanb)) {
          node = (decision_type_[node] & kDefaultLeftMask) ? left_child_[node] : right_child_[node];
        } else {
          node = bin <= threshold_in_bin_[node] ? left_child_[node] : right_child_[node];
        }
      }
    }
    if (is_linear_ && data->has_raw()) {
      const int leaf = ~node;
      double out = leaf_const_.empty() || leaf_coeff_[leaf].empty()
                       ? leaf_value_[leaf] : leaf_const_[leaf];
      if (!leaf_const_.empty() && !leaf_coeff_[leaf].empty()) {
        bool ok = true;
        for (size_t k = 0; k < leaf_coeff_[leaf].size(); ++k) {
          const float v = data->raw_value(leaf_features_inner_[leaf][k], r);
          if (std::isnan(v)) { ok = false; break; }
          out += leaf_coeff_[leaf][k] * v;
        }
        if (!ok) out = leaf_value_[leaf];
      }
      score[r] += out;
    } else {
      score[r] += leaf_value_[~node];
    }
  }
}

}  // namespace migbm
