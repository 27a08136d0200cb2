// cimba_amd statistics: running summaries usable on host and device.
//
// Capability parity with reference src/cmb_datasummary.c (running tally:
// count/mean/min/max/variance/skewness/kurtosis, mergeable across trials)
// and src/cmb_wtdsummary.c (weighted running stats, value x duration, for
// time-averaged metrics).  POD + merge() + raw-sum export so per-GPU
// partials reduce over RCCL/xGMI (the cross-trial merge the
// reference does on the host becomes an RCCL reduce).
#pragma once

#include "config.hpp"
#include <math.h>
#include <float.h>

namespace cmb {

// Running central moments via the one-pass Welford/Pébay update with exact
// pairwise merge — the same capability as cmb_datasummary, shaped so that
// merge() is associative (tree reductions across waves/GPUs are valid).
struct DataSummary {
    double n;     // count (double so the struct reduces as a flat f64 vector)
    double mean;
    double m2, m3, m4;
    double mn, mx;

    CMB_FORCEINLINE void reset() {
        n = 0.0; mean = 0.0; m2 = m3 = m4 = 0.0;
        mn = DBL_MAX; mx = -DBL_MAX;
    }

    CMB_FORCEINLINE void add(double x) {
        const double n1 = n;
        n += 1.0;
        const double delta = x - mean;
        const double dn = delta / n;
        const double dn2 = dn * dn;
        const double t1 = delta * dn * n1;
        mean += dn;
        m4 += t1 * dn2 * (n * n - 3.0 * n + 3.0) + 6.0 * dn2 * m2 - 4.0 * dn * m3;
        m3 += t1 * dn * (n - 2.0) - 3.0 * dn * m2;
        m2 += t1;
        if (x < mn) mn = x;
        if (x > mx) mx = x;
    }

    // Pébay pairwise merge (exact): makes cross-trial / cross-GPU reduction
    // associative.  Counterpart of cmb_datasummary_merge
    // (reference include/cmb_datasummary.h:121).
    CMB_FORCEINLINE void merge(const DataSummary& o) {
        if (o.n == 0.0) return;
        if (n == 0.0) { *this = o; return; }
        const double na = n, nb = o.n, nx = na + nb;
        const double d = o.mean - mean;
        const double d2 = d * d;
        const double m2x = m2 + o.m2 + d2 * na * nb / nx;
        const double m3x = m3 + o.m3 +
            d * d2 * na * nb * (na - nb) / (nx * nx) +
            3.0 * d * (na * o.m2 - nb * m2) / nx;
        const double m4x = m4 + o.m4 +
            d2 * d2 * na * nb * (na * na - na * nb + nb * nb) / (nx * nx * nx) +
            6.0 * d2 * (na * na * o.m2 + nb * nb * m2) / (nx * nx) +
            4.0 * d * (na * o.m3 - nb * m3) / nx;
        mean = (na * mean + nb * o.mean) / nx;
        n = nx; m2 = m2x; m3 = m3x; m4 = m4x;
        if (o.mn < mn) mn = o.mn;
        if (o.mx > mx) mx = o.mx;
    }

    CMB_FORCEINLINE double count() const { return n; }
    CMB_FORCEINLINE double variance() const { return n > 1.0 ? m2 / (n - 1.0) : 0.0; }
    CMB_FORCEINLINE double stddev() const { return sqrt(variance()); }
    CMB_FORCEINLINE double skewness() const {
        return m2 > 0.0 ? sqrt(n) * m3 / pow(m2, 1.5) : 0.0;
    }
    CMB_FORCEINLINE double kurtosis() const {  // excess kurtosis
        return m2 > 0.0 ? n * m4 / (m2 * m2) - 3.0 : 0.0;
    }
};

// Weighted running summary (weight = duration for time-averaged state
// statistics).  Counterpart of cmb_wtdsummary (reference
// src/cmb_wtdsummary.c).
struct WtdSummary {
    double n;        // number of samples
    double sumw;     // total weight
    double mean;     // weighted mean
    double m2;       // weighted sum of squared deviations
    double m3, m4;   // weighted 3rd/4th central sums (skewness/kurtosis)
    double mn, mx;

    CMB_FORCEINLINE void reset() {
        n = 0.0; sumw = 0.0; mean = 0.0; m2 = 0.0; m3 = 0.0; m4 = 0.0;
        mn = DBL_MAX; mx = -DBL_MAX;
    }

    // weighted Pebay update (exact, associative with merge below)
    CMB_FORCEINLINE void add(double x, double w) {
        if (w <= 0.0) return;
        n += 1.0;
        const double wa = sumw, wx = wa + w;
        const double d = x - mean;
        const double dn = d * w / wx;
        const double t1 = d * dn * wa;
        mean += dn;
        m4 += t1 * dn * dn * (wx * wx / (w * w) - 3.0 * wx / w + 3.0) +
              6.0 * dn * dn * m2 - 4.0 * dn * m3;
        m3 += t1 * dn * (wx / w - 2.0) - 3.0 * dn * m2;
        m2 += t1;
        sumw = wx;
        if (x < mn) mn = x;
        if (x > mx) mx = x;
    }

    CMB_FORCEINLINE void merge(const WtdSummary& o) {
        if (o.sumw == 0.0) { n += o.n; return; }
        if (sumw == 0.0) { const double na = n; *this = o; n += na; return; }
        const double wa = sumw, wb = o.sumw, wx = wa + wb;
        const double d = o.mean - mean;
        const double d2 = d * d;
        const double m2x = m2 + o.m2 + d2 * wa * wb / wx;
        const double m3x = m3 + o.m3 +
            d * d2 * wa * wb * (wa - wb) / (wx * wx) +
            3.0 * d * (wa * o.m2 - wb * m2) / wx;
        const double m4x = m4 + o.m4 +
            d2 * d2 * wa * wb * (wa * wa - wa * wb + wb * wb) /
                (wx * wx * wx) +
            6.0 * d2 * (wa * wa * o.m2 + wb * wb * m2) / (wx * wx) +
            4.0 * d * (wa * o.m3 - wb * m3) / wx;
        mean = (wa * mean + wb * o.mean) / wx;
        sumw = wx;
        m2 = m2x; m3 = m3x; m4 = m4x;
        n += o.n;
        if (o.mn < mn) mn = o.mn;
        if (o.mx > mx) mx = o.mx;
    }

    CMB_FORCEINLINE double variance() const { return sumw > 0.0 ? m2 / sumw : 0.0; }
    CMB_FORCEINLINE double stddev() const { return sqrt(variance()); }
    CMB_FORCEINLINE double skewness() const {
        return m2 > 0.0 ? sqrt(sumw) * m3 / pow(m2, 1.5) : 0.0;
    }
    CMB_FORCEINLINE double kurtosis() const {  // excess
        return m2 > 0.0 ? sumw * m4 / (m2 * m2) - 3.0 : 0.0;
    }
};

// Fixed-capacity (value, time) recorder for piecewise-constant state
// histories — the device-capable core of reference cmb_timeseries
// (src/cmb_timeseries.c: (x,t) tuples, time-weighted summarize, finalize at
// end time).  CAP=0 disables recording at compile time.
template <int CAP>
struct TimeseriesRec {
    double x[CAP > 0 ? CAP : 1];
    double t[CAP > 0 ? CAP : 1];
    int32_t len;
    int32_t dropped;

    CMB_FORCEINLINE void reset() { len = 0; dropped = 0; }
    CMB_FORCEINLINE void add(double value, double time) {
        if (CAP <= 0) return;
        if (len < CAP) {
            x[len] = value;
            this->t[len] = time;
            ++len;
        } else {
            ++dropped;
        }
    }
    // time-weighted summary over [t0, end]: each sample holds until the next
    CMB_FORCEINLINE void summarize(WtdSummary& out, double end_time) const {
        for (int32_t i = 0; i < len; ++i) {
            const double t1 = (i + 1 < len) ? t[i + 1] : end_time;
            out.add(x[i], t1 - t[i]);
        }
    }
};

}  // namespace cmb
