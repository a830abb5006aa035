/*!
 * migbm — MI355X-native gradient boosting framework.
 * Core utilities: logging, timing, threading, random.
 *
 * Capability parity target: include/LightGBM/utils/{log.h,common.h,threading.h,random.h}
 * in the reference (see /root/reference). Fresh implementation, not a copy.
 */
#ifndef MIGBM_COMMON_H_
#define MIGBM_COMMON_H_

#include <algorithm>
#include <atomic>
#include <cmath>
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <limits>
#include <memory>
#include <sstream>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#else
inline int omp_get_max_threads() { return 1; }
inline int omp_get_thread_num() { return 0; }
inline void omp_set_num_threads(int) {}
#endif

namespace migbm {

using data_size_t = int32_t;   // row index type (matches reference data_size_t)
using score_t = float;         // gradient/hessian element type
using hist_t = double;         // histogram accumulation type on host
using label_t = float;

constexpr double kEpsilon = 1e-15;
constexpr double kMinScore = -std::numeric_limits<double>::infinity();

// ---------------------------------------------------------------- logging
enum class LogLevel : int { Fatal = -1, Warning = 0, Info = 1, Debug = 2 };

class Log {
 public:
  using Callback = void (*)(const char*);
  static LogLevel& Level() { static LogLevel l = LogLevel::Info; return l; }
  static Callback& Cb() { static Callback cb = nullptr; return cb; }

  static void Write(LogLevel lv, const char* tag, const char* fmt, va_list ap) {
    if (lv > Level()) return;  // the registered callback respects the level too
    char buf[2048];
    vsnprintf(buf, sizeof(buf), fmt, ap);
    char out[2112];
    snprintf(out, sizeof(out), "[migbm] [%s] %s\n", tag, buf);
    if (Cb() != nullptr) {
      Cb()(out);
    } else if (lv <= Level()) {
      fprintf(stderr, "%s", out);
      fflush(stderr);
    }
  }
  static void Debug(const char* fmt, ...) {
    va_list ap; va_start(ap, fmt); Write(LogLevel::Debug, "Debug", fmt, ap); va_end(ap);
  }
  static void Info(const char* fmt, ...) {
    va_list ap; va_start(ap, fmt); Write(LogLevel::Info, "Info", fmt, ap); va_end(ap);
  }
  static void Warning(const char* fmt, ...) {
    va_list ap; va_start(ap, fmt); Write(LogLevel::Warning, "Warning", fmt, ap); va_end(ap);
  }
  [[noreturn]] static void Fatal(const char* fmt, ...) {
    char buf[2048];
    va_list ap; va_start(ap, fmt);
    vsnprintf(buf, sizeof(buf), fmt, ap);
    va_end(ap);
    throw std::runtime_error(std::string("[migbm] [Fatal] ") + buf);
  }
};

#define MIGBM_CHECK(cond) \
  if (!(cond)) migbm::Log::Fatal("Check failed: %s at %s:%d", #cond, __FILE__, __LINE__)
#define MIGBM_CHECK_EQ(a, b) MIGBM_CHECK((a) == (b))
#define MIGBM_CHECK_GT(a, b) MIGBM_CHECK((a) > (b))
#define MIGBM_CHECK_GE(a, b) MIGBM_CHECK((a) >= (b))
#define MIGBM_CHECK_LT(a, b) MIGBM_CHECK((a) < (b))
#define MIGBM_CHECK_LE(a, b) MIGBM_CHECK((a) <= (b))
#define MIGBM_CHECK_NOTNULL(p) \
  if ((p) == nullptr) migbm::Log::Fatal(#p " can't be NULL at %s:%d", __FILE__, __LINE__)

// ---------------------------------------------------------------- random
/*! Deterministic small PRNG (capability: reference utils/random.h). */
class Random {
 public:
  explicit Random(int seed = 0) : x_(0x9E3779B97F4A7C15ULL ^ static_cast<uint64_t>(seed)) {
    // warm up
    for (int i = 0; i < 4; ++i) NextInt64();
  }
  uint64_t NextInt64() {
    // splitmix64
    uint64_t z = (x_ += 0x9E3779B97F4A7C15ULL);
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
    return z ^ (z >> 31);
  }
  /*! random int in [0, n) */
  int NextInt(int lo, int hi) {
    if (hi <= lo) return lo;
    return lo + static_cast<int>(NextInt64() % static_cast<uint64_t>(hi - lo));
  }
  float NextFloat() {
    return static_cast<float>((NextInt64() >> 11) * (1.0 / 9007199254740992.0));
  }
  /*! Sample k of n without replacement, returned sorted ascending. */
  std::vector<int> Sample(int n, int k) {
    std::vector<int> out;
    if (k <= 0 || n <= 0) return out;
    if (k >= n) {
      out.resize(n);
      for (int i = 0; i < n; ++i) out[i] = i;
      return out;
    }
    out.reserve(k);
    // Floyd's algorithm then sort
    std::unordered_map<int, char> seen;
    for (int j = n - k; j < n; ++j) {
      int t = NextInt(0, j + 1);
      if (seen.count(t)) { seen[j] = 1; out.push_back(j); }
      else { seen[t] = 1; out.push_back(t); }
    }
    std::sort(out.begin(), out.end());
    return out;
  }

 private:
  uint64_t x_;
};

// ---------------------------------------------------------------- timer
/*! Phase timer (capability: reference Common::Timer / FunctionTimer). */
class Timer {
 public:
  static Timer& Global() {
    static Timer t;
    static bool reg = [] {
      if (Enabled()) atexit([] { Timer::Global().Print(); });
      return true;
    }();
    (void)reg;
    return t;
  }
  /*! runtime-enabled: set MIGBM_TIMETAG=1; zero cost when off beyond one branch */
  static bool Enabled() {
    static bool e = getenv("MIGBM_TIMETAG") != nullptr;
    return e;
  }
  void Start(const std::string& name) {
    if (Enabled()) starts_[name] = Now();
  }
  void Stop(const std::string& name) {
    if (Enabled()) totals_[name] += Now() - starts_[name];
  }
  void Print() const {
    for (auto& kv : totals_)
      fprintf(stderr, "[timer] %-22s %.3f s\n", kv.first.c_str(), kv.second);
  }
  static double Now() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
  }

 private:
  std::unordered_map<std::string, double> starts_, totals_;
};

// ---------------------------------------------------------------- small linear algebra
/*! Cholesky solve of A beta = b (A: dim x dim, upper triangle filled, SPD after
 *  ridge). Returns false when not positive definite. Shared by the CPU linear
 *  tree fit and the device Gram post-pass. */
inline bool CholeskySolve(const std::vector<double>& A, const std::vector<double>& b,
                          int dim, std::vector<double>* beta_out) {
  std::vector<double> Lm(static_cast<size_t>(dim) * dim, 0.0);
  for (int a = 0; a < dim; ++a) {
    for (int c2 = 0; c2 <= a; ++c2) {
      double sum = A[static_cast<size_t>(std::min(a, c2)) * dim + std::max(a, c2)];
      for (int t = 0; t < c2; ++t) sum -= Lm[a * dim + t] * Lm[c2 * dim + t];
      if (a == c2) {
        if (sum <= 1e-12) return false;
        Lm[a * dim + a] = std::sqrt(sum);
      } else {
        Lm[a * dim + c2] = sum / Lm[c2 * dim + c2];
      }
    }
  }
  std::vector<double> y(dim);
  std::vector<double>& beta = *beta_out;
  beta.assign(dim, 0.0);
  for (int a = 0; a < dim; ++a) {
    double sum = b[a];
    for (int t = 0; t < a; ++t) sum -= Lm[a * dim + t] * y[t];
    y[a] = sum / Lm[a * dim + a];
  }
  for (int a = dim - 1; a >= 0; --a) {
    double sum = y[a];
    for (int t = a + 1; t < dim; ++t) sum -= Lm[t * dim + a] * beta[t];
    beta[a] = sum / Lm[a * dim + a];
  }
  for (double v : beta)
    if (!std::isfinite(v)) return false;
  return true;
}

// ---------------------------------------------------------------- threading
namespace Threading {

/*! Split [0, n) into roughly equal blocks per thread, aligned to `align`. */
template <typename T>
inline void BlockInfo(int num_threads, T n, T align, int* out_nblock, T* block_size) {
  *out_nblock = std::min<int>(num_threads, static_cast<int>((n + align - 1) / align));
  if (*out_nblock > 1) {
    T size = (n + (*out_nblock) - 1) / (*out_nblock);
    *block_size = ((size + align - 1) / align) * align;
  } else {
    *block_size = n;
  }
}

template <typename T, typename F>
inline void For(T start, T end, T min_block, F&& f) {
  int nthread = omp_get_max_threads();
  int nblock;
  T bsize;
  BlockInfo<T>(nthread, end - start, min_block, &nblock, &bsize);
#pragma omp parallel for schedule(static, 1)
  for (int i = 0; i < nblock; ++i) {
    T s = start + static_cast<T>(i) * bsize;
    T e = std::min<T>(end, s + bsize);
    if (s < e) f(i, s, e);
  }
}

}  // namespace Threading

// ---------------------------------------------------------------- string helpers
namespace Common {

inline std::vector<std::string> Split(const char* str, char delim) {
  std::vector<std::string> out;
  const char* p = str;
  const char* s = p;
  while (*p) {
    if (*p == delim) {
      out.emplace_back(s, p - s);
      s = p + 1;
    }
    ++p;
  }
  out.emplace_back(s, p - s);
  return out;
}

inline std::vector<std::string> SplitAny(const char* str, const char* delims) {
  std::vector<std::string> out;
  const char* p = str;
  const char* s = p;
  auto isdelim = [delims](char c) { return strchr(delims, c) != nullptr; };
  while (*p) {
    if (isdelim(*p)) {
      if (p > s) out.emplace_back(s, p - s);
      s = p + 1;
    }
    ++p;
  }
  if (p > s) out.emplace_back(s, p - s);
  return out;
}

inline std::string Trim(const std::string& str) {
  size_t b = str.find_first_not_of(" \t\r\n");
  if (b == std::string::npos) return "";
  size_t e = str.find_last_not_of(" \t\r\n");
  return str.substr(b, e - b + 1);
}

inline std::string ToLower(const std::string& s) {
  std::string r = s;
  std::transform(r.begin(), r.end(), r.begin(), [](char c) { return static_cast<char>(::tolower(c)); });
  return r;
}

inline bool StartsWith(const std::string& s, const std::string& prefix) {
  return s.size() >= prefix.size() && s.compare(0, prefix.size(), prefix) == 0;
}

template <typename T>
inline std::string Join(const std::vector<T>& v, const char* delim) {
  std::stringstream ss;
  for (size_t i = 0; i < v.size(); ++i) {
    if (i) ss << delim;
    ss << v[i];
  }
  return ss.str();
}

/*! Double to shortest round-trip string (model text needs exact round-trip). */
inline std::string DoubleToStr(double x) {
  char buf[64];
  for (int prec = 6; prec <= 17; ++prec) {
    snprintf(buf, sizeof(buf), "%.*g", prec, x);
    if (strtod(buf, nullptr) == x) break;
  }
  return std::string(buf);
}

template <typename T>
inline std::string ArrayToString(const T* arr, size_t n, const char* delim = " ") {
  std::stringstream ss;
  for (size_t i = 0; i < n; ++i) {
    if (i) ss << delim;
    if constexpr (std::is_floating_point<T>::value) ss << DoubleToStr(arr[i]);
    else ss << arr[i];
  }
  return ss.str();
}

inline double Atof(const char* p) { return strtod(p, nullptr); }

template <typename T>
inline void StringToArray(const std::string& s, char delim, std::vector<T>* out) {
  auto toks = Split(s.c_str(), delim);
  out->clear();
  out->reserve(toks.size());
  for (auto& t : toks) {
    if (t.empty()) continue;
    if constexpr (std::is_floating_point<T>::value) out->push_back(static_cast<T>(strtod(t.c_str(), nullptr)));
    else out->push_back(static_cast<T>(strtoll(t.c_str(), nullptr, 10)));
  }
}

/*! Sign-preserving L1 threshold: sign(s) * max(0, |s| - l1).  (gain math) */
inline double ThresholdL1(double s, double l1) {
  const double reg_s = std::max(0.0, std::fabs(s) - l1);
  return s >= 0.0 ? reg_s : -reg_s;
}

inline int RoundInt(double x) { return static_cast<int>(x + 0.5); }

inline double Sigmoid(double x) { return 1.0 / (1.0 + std::exp(-x)); }

/*! AvoidInf as in reference gain math: clamp to +-1e300. */
inline double AvoidInf(double x) {
  if (std::isnan(x)) return 0.0;
  if (x >= 1e300) return 1e300;
  if (x <= -1e300) return -1e300;
  return x;
}

}  // namespace Common

}  // namespace migbm

#endif  // MIGBM_COMMON_H_
