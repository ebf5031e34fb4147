// distref_shim.cpp — C-linkage exports of the reference's own scalar
// distance functions (/root/reference/src/simd/distances_ref.cc, compiled
// from where it lies; never copied).  Used by tests/test_oracle.py to pin
// the oracle's arithmetic core bit-exact.
#include <cstddef>

namespace dingodb {
float fvec_L2sqr_ref(const float* x, const float* y, size_t d);
float fvec_inner_product_ref(const float* x, const float* y, size_t d);
float fvec_norm_L2sqr_ref(const float* x, size_t d);
}  // namespace dingodb

extern "C" {
float ref_fvec_L2sqr(const float* x, const float* y, size_t d) {
  return dingodb::fvec_L2sqr_ref(x, y, d);
}
float ref_fvec_inner_product(const float* x, const float* y, size_t d) {
  return dingodb::fvec_inner_product_ref(x, y, d);
}
float ref_fvec_norm_L2sqr(const float* x, size_t d) {
  return dingodb::fvec_norm_L2sqr_ref(x, d);
}
}
