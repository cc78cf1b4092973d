// Device fp64 truncated-normal math for CDNA4 (kernel K3).
//
// Same tail-stable branch structure as the host reference
// (optuna_amd/samplers/_tpe/_truncnorm_np.py, golden-tested against scipy):
//  * log_ndtr via erfcx in the deep left tail (no underflow),
//  * log_gauss_mass with the left/right/central case split,
//  * ndtri_exp = Newton on log_ndtr from a normcdfinv (or asymptotic) start.
//
// All functions are plain scalar device functions; the TPE kernels inline them.
#pragma once
#include <hip/hip_runtime.h>

namespace tn {

__device__ __forceinline__ double log_ndtr(double x) {
    // log Phi(x), stable over the whole real line.
    if (x > 0.0) {
        // 1 - Phi(-x); erfc keeps precision for moderate x, log1p for x large.
        return log1p(-0.5 * erfc(x * 0.7071067811865475244));
    }
    // Phi(x) = 0.5 * erfcx(-x/sqrt2) * exp(-x^2/2)
    return log(0.5 * erfcx(-x * 0.7071067811865475244)) - 0.5 * x * x;
}

__device__ __forceinline__ double log_norm_pdf(double x) {
    return -0.5 * x * x - 0.9189385332046727418;  // -x^2/2 - log(sqrt(2*pi))
}

// log( Phi(b) - Phi(a) ), a < b.
__device__ __forceinline__ double log_gauss_mass(double a, double b) {
    if (!(a < b)) return -INFINITY;
    if (b <= 0.0) {
        const double log_b = log_ndtr(b);
        return log_b + log1p(-exp(log_ndtr(a) - log_b));
    }
    if (a > 0.0) {
        const double log_a = log_ndtr(-a);
        return log_a + log1p(-exp(log_ndtr(-b) - log_a));
    }
    // Central interval: no cancellation danger in linear space.
    return log1p(-normcdf(a) - normcdf(-b));
}

// Inverse of log_ndtr.
__device__ __forceinline__ double ndtri_exp(double log_p) {
    double x;
    if (log_p > -690.0) {
        const double p = exp(log_p);
        x = normcdfinv(p);
        if (!isfinite(x)) x = (log_p > -1.0) ? 0.0 : -sqrt(-2.0 * log_p);
    } else {
        // Deep left tail: solve -x^2/2 - log(-x) - log(sqrt(2*pi)) = log_p.
        x = -sqrt(-2.0 * log_p);
        for (int i = 0; i < 3; ++i) {
            x = -sqrt(-2.0 * (log_p + log(-x) + 0.9189385332046727418));
        }
    }
    // Newton refinement on f(x) = log_ndtr(x) - log_p; f'(x) = phi(x)/Phi(x).
    for (int i = 0; i < 3; ++i) {
        const double f = log_ndtr(x) - log_p;
        const double fp = exp(log_norm_pdf(x) - log_ndtr(x));
        if (fp <= 0.0 || !isfinite(fp)) break;
        const double step = f / fp;
        x -= step;
        if (fabs(step) < 1e-14 * (1.0 + fabs(x))) break;
    }
    return x;
}

// Quantile of the standard normal truncated to [a, b]; q in [0, 1].
__device__ __forceinline__ double trunc_ppf(double q, double a, double b) {
    if (a == b) return a;
    double x;
    if (a < 0.0) {
        // log Phi(x) = logaddexp(log Phi(a), log q + log mass)
        const double la = log_ndtr(a);
        const double lm = (q > 0.0) ? log(q) + log_gauss_mass(a, b) : -INFINITY;
        const double m = fmax(la, lm);
        const double log_phi_x =
            (m == -INFINITY) ? -INFINITY : m + log(exp(la - m) + exp(lm - m));
        x = ndtri_exp(log_phi_x);
    } else {
        const double lb = log_ndtr(-b);
        const double lm = (q < 1.0) ? log1p(-q) + log_gauss_mass(a, b) : -INFINITY;
        const double m = fmax(lb, lm);
        const double log_sf_x =
            (m == -INFINITY) ? -INFINITY : m + log(exp(lb - m) + exp(lm - m));
        x = -ndtri_exp(log_sf_x);
    }
    return fmin(fmax(x, a), b);
}

// log pdf of normal(loc, scale) truncated to [loc+a*scale, loc+b*scale] at x.
__device__ __forceinline__ double trunc_logpdf(double x, double a, double b,
                                               double loc, double scale) {
    const double z = (x - loc) / scale;
    if (z < a || z > b) return -INFINITY;
    return log_norm_pdf(z) - log_gauss_mass(a, b) - log(scale);
}

}  // namespace tn
