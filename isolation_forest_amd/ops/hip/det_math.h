// Deterministic transcendentals — EXACT mirror of utils/det_math.py.
// Only IEEE-correctly-rounded double ops (+,-,*,/ , sqrt, frexp, floor) in a
// fixed evaluation order, so device doubles are bit-identical to numpy's.
#pragma once

#include <hip/hip_runtime.h>

namespace ifa {

__device__ __forceinline__ double det_log(double x) {
  // x in (0, 1]; x = m * 2^e with m in [0.5, 1)
  int e;
  double m = frexp(x, &e);
  double z = (m - 1.0) / (m + 1.0);
  double z2 = z * z;
  // atanh series coeffs 2/(2k+1), k=10..0 (Horner), same order as numpy
  double s = 2.0 / 21.0;
  s = s * z2 + 2.0 / 19.0;
  s = s * z2 + 2.0 / 17.0;
  s = s * z2 + 2.0 / 15.0;
  s = s * z2 + 2.0 / 13.0;
  s = s * z2 + 2.0 / 11.0;
  s = s * z2 + 2.0 / 9.0;
  s = s * z2 + 2.0 / 7.0;
  s = s * z2 + 2.0 / 5.0;
  s = s * z2 + 2.0 / 3.0;
  s = s * z2 + 2.0;
  s = s * z;
  const double LN2 = 6.93147180559945286227e-01;
  return s + (double)e * LN2;
}

__device__ __forceinline__ double det_poly_cos(double a2) {
  // sum (-1)^k a^(2k)/(2k)!, k=8..0
  double s = 1.0 / 20922789888000.0;   //  1/16!
  s = s * a2 + -1.0 / 87178291200.0;   // -1/14!
  s = s * a2 + 1.0 / 479001600.0;      //  1/12!
  s = s * a2 + -1.0 / 3628800.0;       // -1/10!
  s = s * a2 + 1.0 / 40320.0;          //  1/8!
  s = s * a2 + -1.0 / 720.0;           // -1/6!
  s = s * a2 + 1.0 / 24.0;             //  1/4!
  s = s * a2 + -1.0 / 2.0;             // -1/2!
  s = s * a2 + 1.0;
  return s;
}

__device__ __forceinline__ double det_poly_sin(double a2) {
  double s = 1.0 / 355687428096000.0;   //  1/17!
  s = s * a2 + -1.0 / 1307674368000.0;  // -1/15!
  s = s * a2 + 1.0 / 6227020800.0;      //  1/13!
  s = s * a2 + -1.0 / 39916800.0;       // -1/11!
  s = s * a2 + 1.0 / 362880.0;          //  1/9!
  s = s * a2 + -1.0 / 5040.0;           // -1/7!
  s = s * a2 + 1.0 / 120.0;             //  1/5!
  s = s * a2 + -1.0 / 6.0;              // -1/3!
  s = s * a2 + 1.0;
  return s;
}

__device__ __forceinline__ double det_cos2pi(double u) {
  const double PI_2 = 1.57079632679489661923;
  double s4 = u * 4.0;
  double q = floor(s4);
  double f = s4 - q;
  long qi = (long)q & 3;
  bool use_sin_half = f > 0.5;
  double g = use_sin_half ? (1.0 - f) : f;
  double a = g * PI_2;
  double a2 = a * a;
  double cosv = det_poly_cos(a2);
  double sinv = det_poly_sin(a2) * a;
  double cos_q = use_sin_half ? sinv : cosv;
  double sin_q = use_sin_half ? cosv : sinv;
  if (qi == 0) return cos_q;
  if (qi == 1) return -sin_q;
  if (qi == 2) return -cos_q;
  return sin_q;
}

__device__ __forceinline__ double det_gaussian(double u1, double u2) {
  double r = 0.0;
  if (u1 < 1.0) {
    double t = 1.0 - u1;
    if (t < 1e-300) t = 1e-300;
    // det_log(1.0) is ~1e-12, not exactly 0 (polynomial residual), so the
    // u1 == 0 draw would take sqrt of a negative and poison the node's
    // weights with NaN (fuzz-found at 2^-24 probability per draw). The
    // true Box-Muller radial term at u1 = 0 is 0: clamp the argument.
    double a = -2.0 * det_log(t);
    r = sqrt(a > 0.0 ? a : 0.0);
  }
  return r * det_cos2pi(u2);
}

}  // namespace ifa
