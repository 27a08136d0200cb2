// Host-side sample container + time series — capability parity with
// reference src/cmb_dataset.c (growable array of doubles, non-recursive
// heapsort, median/five-number summary, histogram, ACF + PACF via
// Durbin-Levinson, correlogram, copy/merge/summarize; cmb_dataset.c:161,
// :697-900) and src/cmb_timeseries.c ((x,t) tuples for piecewise-constant
// state histories, time-weighted summarize, finalize at end time).
//
// Host-only (std::vector); the device-capable fixed-capacity recorder is
// TimeseriesRec in stats.hpp.  Statistics meanings match the reference;
// the implementation is original (heapsort kept non-recursive like the
// reference's for identical in-place behavior on ties).
#pragma once

#include "stats.hpp"

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <vector>

namespace cmb {

class Dataset {
  public:
    void add(double x) { xs_.push_back(x); sorted_ = false; }
    size_t size() const { return xs_.size(); }
    const std::vector<double>& values() const { return xs_; }
    void clear() { xs_.clear(); sorted_ = true; }

    void merge(const Dataset& o) {
        xs_.insert(xs_.end(), o.xs_.begin(), o.xs_.end());
        sorted_ = false;
    }

    // non-recursive heapsort (reference cmb_dataset.c:161)
    void sort() {
        if (sorted_) return;
        const int64_t n = (int64_t)xs_.size();
        for (int64_t i = n / 2 - 1; i >= 0; --i) sift_(i, n);
        for (int64_t i = n - 1; i > 0; --i) {
            std::swap(xs_[0], xs_[i]);
            sift_(0, i);
        }
        sorted_ = true;
    }

    double quantile(double q) {
        sort();
        if (xs_.empty()) return 0.0;
        const double pos = q * (double)(xs_.size() - 1);
        const size_t lo = (size_t)pos;
        const double frac = pos - (double)lo;
        if (lo + 1 >= xs_.size()) return xs_.back();
        return xs_[lo] * (1.0 - frac) + xs_[lo + 1] * frac;
    }
    double median() { return quantile(0.5); }

    // min, Q1, median, Q3, max (reference fivenum_print)
    void fivenum(double out[5]) {
        sort();
        out[0] = xs_.empty() ? 0.0 : xs_.front();
        out[1] = quantile(0.25);
        out[2] = quantile(0.5);
        out[3] = quantile(0.75);
        out[4] = xs_.empty() ? 0.0 : xs_.back();
    }

    DataSummary summarize() const {
        DataSummary s;
        s.reset();
        for (double x : xs_) s.add(x);
        return s;
    }

    // equal-width histogram over [min, max]
    std::vector<int64_t> histogram(int nbins) {
        sort();
        std::vector<int64_t> h((size_t)nbins, 0);
        if (xs_.empty() || nbins <= 0) return h;
        const double lo = xs_.front(), hi = xs_.back();
        const double w = hi > lo ? (hi - lo) / nbins : 1.0;
        for (double x : xs_) {
            int b = (int)((x - lo) / w);
            if (b >= nbins) b = nbins - 1;
            if (b < 0) b = 0;
            h[(size_t)b]++;
        }
        return h;
    }

    // autocorrelation function r_1..r_maxlag (reference ACF,
    // cmb_dataset.c:697-900)
    std::vector<double> acf(int maxlag) const {
        const size_t n = xs_.size();
        std::vector<double> r((size_t)maxlag, 0.0);
        if (n < 2) return r;
        double mu = 0.0;
        for (double x : xs_) mu += x;
        mu /= (double)n;
        double c0 = 0.0;
        for (double x : xs_) c0 += (x - mu) * (x - mu);
        if (c0 <= 0.0) return r;
        for (int k = 1; k <= maxlag && (size_t)k < n; ++k) {
            double ck = 0.0;
            for (size_t t = 0; t + (size_t)k < n; ++t)
                ck += (xs_[t] - mu) * (xs_[t + (size_t)k] - mu);
            r[(size_t)k - 1] = ck / c0;
        }
        return r;
    }

    // partial autocorrelation via Durbin-Levinson recursion (reference
    // PACF, cmb_dataset.c:697-900)
    std::vector<double> pacf(int maxlag) const {
        std::vector<double> r = acf(maxlag);
        std::vector<double> p((size_t)maxlag, 0.0);
        if (r.empty()) return p;
        std::vector<double> phi_prev((size_t)maxlag + 1, 0.0);
        std::vector<double> phi((size_t)maxlag + 1, 0.0);
        double v = 1.0;
        for (int k = 1; k <= maxlag; ++k) {
            double num = r[(size_t)k - 1];
            for (int j = 1; j < k; ++j)
                num -= phi_prev[(size_t)j] * r[(size_t)(k - j) - 1];
            const double a = v > 0.0 ? num / v : 0.0;
            phi[(size_t)k] = a;
            for (int j = 1; j < k; ++j)
                phi[(size_t)j] =
                    phi_prev[(size_t)j] - a * phi_prev[(size_t)(k - j)];
            v *= (1.0 - a * a);
            p[(size_t)k - 1] = a;
            phi_prev = phi;
        }
        return p;
    }

  private:
    void sift_(int64_t root, int64_t n) {
        double x = xs_[(size_t)root];
        for (;;) {
            int64_t c = 2 * root + 1;
            if (c >= n) break;
            if (c + 1 < n && xs_[(size_t)c + 1] > xs_[(size_t)c]) ++c;
            if (xs_[(size_t)c] <= x) break;
            xs_[(size_t)root] = xs_[(size_t)c];
            root = c;
        }
        xs_[(size_t)root] = x;
    }

    std::vector<double> xs_;
    bool sorted_ = true;
};

// (value, time) series for piecewise-constant state histories
class Timeseries {
  public:
    void add(double x, double t) { xs_.push_back(x); ts_.push_back(t); }
    size_t size() const { return xs_.size(); }
    const std::vector<double>& values() const { return xs_; }
    const std::vector<double>& times() const { return ts_; }

    // each sample holds until the next; the last holds until end_time
    WtdSummary summarize(double end_time) const {
        WtdSummary s;
        s.reset();
        for (size_t i = 0; i < xs_.size(); ++i) {
            const double t1 = (i + 1 < ts_.size()) ? ts_[i + 1] : end_time;
            s.add(xs_[i], t1 - ts_[i]);
        }
        return s;
    }

    // time-weighted median of the state value
    double median(double end_time) const {
        std::vector<std::pair<double, double>> vw;
        double tot = 0.0;
        for (size_t i = 0; i < xs_.size(); ++i) {
            const double t1 = (i + 1 < ts_.size()) ? ts_[i + 1] : end_time;
            const double w = t1 - ts_[i];
            if (w > 0) {
                vw.emplace_back(xs_[i], w);
                tot += w;
            }
        }
        std::sort(vw.begin(), vw.end());
        double acc = 0.0;
        for (auto& p : vw) {
            acc += p.second;
            if (acc >= 0.5 * tot) return p.first;
        }
        return vw.empty() ? 0.0 : vw.back().first;
    }

  private:
    std::vector<double> xs_, ts_;
};

}  // namespace cmb
