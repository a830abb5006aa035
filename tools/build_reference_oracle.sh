#!/bin/bash
# Build the reference LightGBM (mounted read-only at /root/reference) as a CPU-only
# cross-validation oracle for tests/test_reference_compat.py.
# The reference snapshot ships with UNPOPULATED submodules (fast_double_parser, fmt,
# eigen); minimal test-harness shims (strtod / snprintf round-trip / Gauss-Jordan)
# are generated here so the oracle compiles. Shims live in /tmp, not in this repo.
set -e
SHIM=/tmp/lgbshim
BUILD=/tmp/lgbref
SRC=/tmp/lgbrefsrc
OUT="$(cd "$(dirname "$0")" && pwd)/oracle"
# /root/reference must stay pristine: LightGBM's CMake drops lib_lightgbm.so into
# the SOURCE root, so build from a disposable copy of the tree.
mkdir -p $SHIM/fmt $SHIM/Eigen $BUILD "$OUT"
rm -rf $SRC
cp -r /root/reference $SRC

cat > $SHIM/fast_double_parser.h <<'EOF'
#pragma once
#include <cstdlib>
namespace fast_double_parser {
inline const char* parse_number(const char* p, double* out) {
  char* end = nullptr; *out = std::strtod(p, &end);
  return end == p ? nullptr : end;
}
}
EOF

cat > $SHIM/fmt/format.h <<'EOF'
#pragma once
#include <cstdio>
#include <cstring>
#include <string>
#include <type_traits>
namespace fmt {
struct format_to_n_result_t { size_t size; };
template <typename T>
inline format_to_n_result_t format_to_n(char* buf, size_t len, const char* spec, T value) {
  int n = 0;
  if (std::strcmp(spec, "{:g}") == 0) n = snprintf(buf, len, "%g", static_cast<double>(value));
  else if (std::strcmp(spec, "{:.17g}") == 0) n = snprintf(buf, len, "%.17g", static_cast<double>(value));
  else {
    if constexpr (std::is_floating_point<T>::value) {
      for (int prec = 6; prec <= 17; ++prec) {
        n = snprintf(buf, len, "%.*g", prec, static_cast<double>(value));
        if (strtod(buf, nullptr) == static_cast<double>(value)) break;
      }
    } else if constexpr (std::is_signed<T>::value) {
      n = snprintf(buf, len, "%lld", static_cast<long long>(value));
    } else {
      n = snprintf(buf, len, "%llu", static_cast<unsigned long long>(value));
    }
  }
  return {static_cast<size_t>(n < 0 ? len : static_cast<size_t>(n))};
}
}
EOF

cp "$(dirname "$0")/eigen_shim.h" $SHIM/Eigen/Dense

cd $BUILD
cmake $SRC -DCMAKE_BUILD_TYPE=Release -DCMAKE_CXX_FLAGS="-I$SHIM" > /dev/null
make -j16 2>/dev/null || make -j16
cp $SRC/lib_lightgbm.so "$OUT/lib_lightgbm.so"
echo "oracle ready: $OUT/lib_lightgbm.so"
