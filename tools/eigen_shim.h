// Test-harness Eigen shim: the minimal MatrixXd surface linear_tree_learner.cpp uses.
#pragma once
#include <algorithm>
#include <cmath>
#include <cstddef>
#include <vector>
namespace Eigen {
class MatrixXd;
namespace detail { MatrixXd InverseOf(const MatrixXd& a); }
class MatrixXd {
 public:
  MatrixXd() : r_(0), c_(0) {}
  MatrixXd(size_t r, size_t c) : r_(r), c_(c), d_(r * c, 0.0) {}
  template <typename I, typename J>
  double& operator()(I i, J j) { return d_[static_cast<size_t>(i) * c_ + static_cast<size_t>(j)]; }
  template <typename I, typename J>
  double operator()(I i, J j) const { return d_[static_cast<size_t>(i) * c_ + static_cast<size_t>(j)]; }
  template <typename I>
  double& operator()(I i) { return d_[static_cast<size_t>(i)]; }  // vector-style access
  template <typename I>
  double operator()(I i) const { return d_[static_cast<size_t>(i)]; }
  size_t rows() const { return r_; }
  size_t cols() const { return c_; }
  struct LU {
    const MatrixXd* a;
    MatrixXd inverse() const { return detail::InverseOf(*a); }
  };
  LU fullPivLu() const { return LU{this}; }
  MatrixXd operator*(const MatrixXd& o) const {
    MatrixXd out(r_, o.c_);
    for (size_t i = 0; i < r_; ++i)
      for (size_t k = 0; k < c_; ++k) {
        double v = (*this)(i, k);
        if (v == 0.0) continue;
        for (size_t j = 0; j < o.c_; ++j) out(i, j) += v * o(k, j);
      }
    return out;
  }
  MatrixXd operator-() const {
    MatrixXd out = *this;
    for (auto& v : out.d_) v = -v;
    return out;
  }
  std::vector<double> d_;

 private:
  size_t r_, c_;
};
namespace detail {
inline MatrixXd InverseOf(const MatrixXd& a) {
  const size_t n = a.rows();
  MatrixXd m = a, inv(n, n);
  for (size_t i = 0; i < n; ++i) inv(i, i) = 1.0;
  for (size_t col = 0; col < n; ++col) {
    size_t piv = col;
    for (size_t r2 = col + 1; r2 < n; ++r2)
      if (std::abs(m(r2, col)) > std::abs(m(piv, col))) piv = r2;
    if (m(piv, col) == 0.0) continue;
    for (size_t j = 0; j < n; ++j) {
      std::swap(m(piv, j), m(col, j));
      std::swap(inv(piv, j), inv(col, j));
    }
    const double p = m(col, col);
    for (size_t j = 0; j < n; ++j) { m(col, j) /= p; inv(col, j) /= p; }
    for (size_t r2 = 0; r2 < n; ++r2) {
      if (r2 == col) continue;
      const double f = m(r2, col);
      if (f == 0.0) continue;
      for (size_t j = 0; j < n; ++j) {
        m(r2, j) -= f * m(col, j);
        inv(r2, j) -= f * inv(col, j);
      }
    }
  }
  return inv;
}
}  // namespace detail
}  // namespace Eigen
