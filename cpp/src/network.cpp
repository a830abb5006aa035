/*! migbm Network implementation over an injected allgather collective. */
#include "migbm/network.h"

namespace migbm {

int Network::num_machines_ = 1;
int Network::rank_ = 0;
AllgatherFn Network::allgather_ = nullptr;

void Network::Init(int num_machines, int rank, AllgatherFn allgather) {
  num_machines_ = num_machines;
  rank_ = rank;
  allgather_ = allgather;
  Log::Info("Network initialized: rank %d / %d machines", rank, num_machines);
}

void Network::Free() {
  num_machines_ = 1;
  rank_ = 0;
  allgather_ = nullptr;
}

void Network::Allgather(const char* input, int size, char* output) {
  if (num_machines_ <= 1) {
    memcpy(output, input, size);
    return;
  }
  MIGBM_CHECK_NOTNULL(allgather_);
  allgather_(input, size, output);
}

void Network::AllgatherV(const char* input, int my_size, const int* sizes, char* output) {
  // pad to max size, gather, compact
  int max_size = 0;
  for (int i = 0; i < num_machines_; ++i) max_size = std::max(max_size, sizes[i]);
  std::vector<char> in_pad(max_size, 0), out_pad(static_cast<size_t>(max_size) * num_machines_);
  memcpy(in_pad.data(), input, my_size);
  Allgather(in_pad.data(), max_size, out_pad.data());
  char* dst = output;
  for (int i = 0; i < num_machines_; ++i) {
    memcpy(dst, out_pad.data() + static_cast<size_t>(i) * max_size, sizes[i]);
    dst += sizes[i];
  }
}

void Network::AllreduceSum(double* data, size_t n) {
  if (num_machines_ <= 1) return;
  // chunked gather+local-sum (correctness path; RCCL handles the hot GPU path)
  const size_t chunk = 1 << 20;
  std::vector<char> out;
  for (size_t off = 0; off < n; off += chunk) {
    size_t m = std::min(chunk, n - off);
    out.resize(m * sizeof(double) * num_machines_);
    Allgather(reinterpret_cast<const char*>(data + off), static_cast<int>(m * sizeof(double)),
              out.data());
    const double* gathered = reinterpret_cast<const double*>(out.data());
    for (size_t i = 0; i < m; ++i) {
      double s = 0;
      for (int r = 0; r < num_machines_; ++r) s += gathered[static_cast<size_t>(r) * m + i];
      data[off + i] = s;
    }
  }
}

void Network::AllreduceSum(float* data, size_t n) {
  if (num_machines_ <= 1) return;
  const size_t chunk = 1 << 20;
  std::vector<char> out;
  for (size_t off = 0; off < n; off += chunk) {
    size_t m = std::min(chunk, n - off);
    out.resize(m * sizeof(float) * num_machines_);
    Allgather(reinterpret_cast<const char*>(data + off), static_cast<int>(m * sizeof(float)),
              out.data());
    const float* gathered = reinterpret_cast<const float*>(out.data());
    for (size_t i = 0; i < m; ++i) {
      float s = 0;
      for (int r = 0; r < num_machines_; ++r) s += gathered[static_cast<size_t>(r) * m + i];
      data[off + i] = s;
    }
  }
}

void Network::AllreduceSum(int64_t* data, size_t n) {
  if (num_machines_ <= 1) return;
  std::vector<char> out(n * sizeof(int64_t) * num_machines_);
  Allgather(reinterpret_cast<const char*>(data), static_cast<int>(n * sizeof(int64_t)),
            out.data());
  const int64_t* gathered = reinterpret_cast<const int64_t*>(out.data());
  for (size_t i = 0; i < n; ++i) {
    int64_t s = 0;
    for (int r = 0; r < num_machines_; ++r) s += gathered[static_cast<size_t>(r) * n + i];
    data[i] = s;
  }
}

double Network::GlobalSyncUpByMean(double local) {
  if (num_machines_ <= 1) return local;
  double v = local;
  AllreduceSum(&v, 1);
  return v / num_machines_;
}

double Network::GlobalSyncUpBySum(double local) {
  if (num_machines_ <= 1) return local;
  double v = local;
  AllreduceSum(&v, 1);
  return v;
}

int64_t Network::GlobalSyncUpBySum(int64_t local) {
  if (num_machines_ <= 1) return local;
  int64_t v = local;
  AllreduceSum(&v, 1);
  return v;
}

}  // namespace migbm
