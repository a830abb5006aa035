/*! migbm DatasetLoader — CSV/TSV/LibSVM text loading with label/weight/query handling and
 *  distributed row sharding. Parity target: reference src/io/dataset_loader.cpp +
 *  parser.cpp (autodetect), metadata.cpp (.weight/.query sidecar files). */
#include "migbm/dataset.h"
#include "migbm/network.h"

#include <cstdio>
#include <cstring>

namespace migbm {

namespace {

enum class FileFormat { kCSV, kTSV, kLibSVM, kSpace };

FileFormat DetectFormat(const std::string& first_line) {
  if (first_line.find(':') != std::string::npos &&
      first_line.find_first_of(",\t") == std::string::npos)
    return FileFormat::kLibSVM;
  if (first_line.find('\t') != std::string::npos) return FileFormat::kTSV;
  if (first_line.find(',') != std::string::npos) return FileFormat::kCSV;
  return FileFormat::kSpace;
}

bool ReadLines(const char* filename, std::vector<std::string>* lines) {
  FILE* fp = fopen(filename, "rb");
  if (!fp) return false;
  fseek(fp, 0, SEEK_END);
  long sz = ftell(fp);
  fseek(fp, 0, SEEK_SET);
  std::string buf(sz, '\0');
  if (fread(&buf[0], 1, sz, fp) != static_cast<size_t>(sz)) { fclose(fp); return false; }
  fclose(fp);
  size_t start = 0;
  for (size_t i = 0; i <= buf.size(); ++i) {
    if (i == buf.size() || buf[i] == '\n') {
      size_t end = i;
      if (end > start && buf[end - 1] == '\r') --end;
      if (end > start) lines->emplace_back(buf.substr(start, end - start));
      start = i + 1;
    }
  }
  return true;
}

int ParseLabelIdx(const Config& cfg) {
  if (cfg.label_column.empty()) return 0;
  if (Common::StartsWith(cfg.label_column, "name:")) return 0;  // name-resolved later
  return atoi(cfg.label_column.c_str());
}

/*! resolve a column spec ("3" or "name:foo" against a header line) to an index;
 *  -1 = unset */
int ResolveColumn(const std::string& spec, const std::vector<std::string>& header) {
  if (spec.empty()) return -1;
  if (Common::StartsWith(spec, "name:")) {
    const std::string want = spec.substr(5);
    for (size_t i = 0; i < header.size(); ++i)
      if (header[i] == want) return static_cast<int>(i);
    Log::Fatal("Column '%s' not found in the header", want.c_str());
  }
  return atoi(spec.c_str());
}

}  // namespace

void DatasetLoader::ParseFile(const char* filename, std::vector<std::vector<double>>* rows,
                              std::vector<float>* labels, std::vector<float>* weights,
                              std::vector<int32_t>* groups, int* out_ncol, int rank,
                              int num_machines) {
  std::vector<std::string> lines;
  if (!ReadLines(filename, &lines)) Log::Fatal("Cannot open data file %s", filename);
  if (lines.empty()) Log::Fatal("Data file %s is empty", filename);
  size_t first = 0;
  if (cfg_.header) first = 1;
  FileFormat fmt = DetectFormat(lines[first]);
  const char* delims = fmt == FileFormat::kCSV ? "," : (fmt == FileFormat::kTSV ? "\t" : " ");
  // column specs (weight_column / group_column / ignore_column; "name:" needs header)
  std::vector<std::string> header_toks;
  if (cfg_.header) header_toks = Common::SplitAny(lines[0].c_str(), delims);
  int label_idx = ParseLabelIdx(cfg_);
  if (Common::StartsWith(cfg_.label_column, "name:"))
    label_idx = ResolveColumn(cfg_.label_column, header_toks);
  const int weight_idx = ResolveColumn(cfg_.weight_column, header_toks);
  const int group_idx = ResolveColumn(cfg_.group_column, header_toks);
  std::vector<int8_t> ignored;
  if (!cfg_.ignore_column.empty()) {
    for (auto& tok : Common::Split(cfg_.ignore_column.c_str(), ',')) {
      auto t = Common::Trim(tok);
      if (t.empty()) continue;
      const int idx = ResolveColumn(t, header_toks);
      if (idx >= 0) {
        if (idx >= static_cast<int>(ignored.size())) ignored.resize(idx + 1, 0);
        ignored[idx] = 1;
      }
    }
  }
  std::vector<int32_t> row_group_ids;
  int ncol = -1;
  const size_t n_lines = lines.size();
  for (size_t li = first; li < n_lines; ++li) {
    // distributed round-robin sharding when pre_partition is off
    if (num_machines > 1 && static_cast<int>((li - first) % num_machines) != rank) continue;
    auto toks = Common::SplitAny(lines[li].c_str(), delims);
    if (toks.empty()) continue;
    std::vector<double> row;
    float label = 0;
    if (fmt == FileFormat::kLibSVM) {
      label = static_cast<float>(atof(toks[0].c_str()));
      for (size_t t = 1; t < toks.size(); ++t) {
        auto kv = Common::Split(toks[t].c_str(), ':');
        if (kv.size() != 2) continue;
        int idx = atoi(kv[0].c_str());
        if (idx >= static_cast<int>(row.size()))
          row.resize(idx + 1, 0.0);
        row[idx] = atof(kv[1].c_str());
      }
    } else {
      row.reserve(toks.size() - 1);
      for (size_t t = 0; t < toks.size(); ++t) {
        const int ti = static_cast<int>(t);
        if (ti == label_idx) {
          label = static_cast<float>(atof(toks[t].c_str()));
        } else if (ti == weight_idx) {
          weights->push_back(static_cast<float>(atof(toks[t].c_str())));
        } else if (ti == group_idx) {
          row_group_ids.push_back(atoi(toks[t].c_str()));
        } else if (ti < static_cast<int>(ignored.size()) && ignored[ti]) {
          // dropped
        } else {
          const std::string& s = toks[t];
          if (s.empty() || s == "na" || s == "NA" || s == "nan" || s == "NaN" || s == "?")
            row.push_back(std::numeric_limits<double>::quiet_NaN());
          else
            row.push_back(atof(s.c_str()));
        }
      }
    }
    ncol = std::max<int>(ncol, static_cast<int>(row.size()));
    rows->push_back(std::move(row));
    labels->push_back(label);
  }
  // group column: run-length encode consecutive equal ids into group sizes
  if (!row_group_ids.empty() && groups->empty()) {
    int32_t cur = row_group_ids[0], cnt = 0;
    for (int32_t g : row_group_ids) {
      if (g == cur) {
        ++cnt;
      } else {
        groups->push_back(cnt);
        cur = g;
        cnt = 1;
      }
    }
    groups->push_back(cnt);
  }
  // pad jagged libsvm rows
  for (auto& r : *rows) r.resize(ncol, 0.0);
  *out_ncol = ncol;

  // sidecar .weight / .query files
  std::string wf = std::string(filename) + ".weight";
  std::vector<std::string> wl;
  if (ReadLines(wf.c_str(), &wl)) {
    for (auto& l : wl) weights->push_back(static_cast<float>(atof(l.c_str())));
  }
  std::string qf = std::string(filename) + ".query";
  std::vector<std::string> ql;
  if (ReadLines(qf.c_str(), &ql)) {
    for (auto& l : ql) groups->push_back(atoi(l.c_str()));
  }
}

/*! .position sidecar (position-debiased ranking): one integer per row. */
static std::vector<int32_t> ReadPositionSidecar(const char* filename) {
  std::vector<int32_t> positions;
  std::vector<std::string> pl;
  if (ReadLines((std::string(filename) + ".position").c_str(), &pl)) {
    for (auto& l : pl) positions.push_back(atoi(l.c_str()));
  }
  return positions;
}

std::unique_ptr<Dataset> DatasetLoader::LoadFromFile(const char* filename, int rank,
                                                     int num_machines) {
  std::vector<std::vector<double>> rows;
  std::vector<float> labels, weights;
  std::vector<int32_t> groups;
  int ncol = 0;
  ParseFile(filename, &rows, &labels, &weights, &groups, &ncol, rank, num_machines);
  auto d = std::make_unique<Dataset>();
  auto at = [&rows](data_size_t r, int c) { return rows[r][c]; };
  std::vector<int8_t> cat_flags;
  {
    Config tmp = cfg_;
    cat_flags.assign(ncol, 0);
    if (!cfg_.categorical_feature.empty()) {
      for (auto& tok : Common::Split(cfg_.categorical_feature.c_str(), ',')) {
        auto t = Common::Trim(tok);
        if (t.empty()) continue;
        int idx = atoi(t.c_str());
        if (idx >= 0 && idx < ncol) cat_flags[idx] = 1;
      }
    }
  }
  d->ConstructFromMat(at, static_cast<data_size_t>(rows.size()), ncol, cfg_, cat_flags);
  d->metadata().SetLabel(labels.data(), static_cast<data_size_t>(labels.size()));
  if (!weights.empty())
    d->metadata().SetWeights(weights.data(), static_cast<data_size_t>(weights.size()));
  if (!groups.empty())
    d->metadata().SetQuery(groups.data(), static_cast<data_size_t>(groups.size()));
  {
    auto positions = ReadPositionSidecar(filename);
    if (!positions.empty())
      d->metadata().SetPosition(positions.data(), static_cast<data_size_t>(positions.size()));
  }
  return d;
}

std::unique_ptr<Dataset> DatasetLoader::LoadFromFileAlignWithOtherDataset(
    const char* filename, const Dataset* train) {
  std::vector<std::vector<double>> rows;
  std::vector<float> labels, weights;
  std::vector<int32_t> groups;
  int ncol = 0;
  ParseFile(filename, &rows, &labels, &weights, &groups, &ncol, 0, 1);
  auto at = [&rows, ncol](data_size_t r, int c) {
    return c < ncol ? rows[r][c] : 0.0;
  };
  auto d = train->CreateValid(at, static_cast<data_size_t>(rows.size()));
  d->metadata().SetLabel(labels.data(), static_cast<data_size_t>(labels.size()));
  if (!weights.empty())
    d->metadata().SetWeights(weights.data(), static_cast<data_size_t>(weights.size()));
  if (!groups.empty())
    d->metadata().SetQuery(groups.data(), static_cast<data_size_t>(groups.size()));
  {
    auto positions = ReadPositionSidecar(filename);
    if (!positions.empty())
      d->metadata().SetPosition(positions.data(), static_cast<data_size_t>(positions.size()));
  }
  return d;
}

/*! Raw-row loading used by LGBM_BoosterPredictForFile. */
std::vector<std::vector<double>> LoadRawRowsForPredict(const char* filename, const Config& cfg,
                                                       int expected_ncol) {
  DatasetLoader loader(cfg);
  std::vector<std::vector<double>> rows;
  std::vector<float> labels, weights;
  std::vector<int32_t> groups;
  int ncol = 0;
  // reuse the parser through a throwaway loader call path
  struct Access : DatasetLoader {
    using DatasetLoader::DatasetLoader;
    void Call(const char* f, std::vector<std::vector<double>>* r, std::vector<float>* l,
              std::vector<float>* w, std::vector<int32_t>* g, int* nc) {
      ParseFile(f, r, l, w, g, nc, 0, 1);
    }
  };
  Access a(cfg);
  a.Call(filename, &rows, &labels, &weights, &groups, &ncol);
  for (auto& r : rows) r.resize(expected_ncol, 0.0);
  return rows;
}

}  // namespace migbm
