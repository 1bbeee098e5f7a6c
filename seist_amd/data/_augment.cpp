// K20 step 2: the full per-sample preprocessing pipeline in C++ —
// noise gating, phase padding, the 11-augmentation chain in the EXACT
// np.random draw order of the Python path (data/preprocess.py, itself
// draw-order-exact with the reference training/preprocess.py), window
// cutting and normalization. Draws come from the bit-exact RandomState
// port (_rng.h), so with identical seeding the output is bit-identical
// to the numpy path (tests/test_preprocess.py::test_native_process_*).
//
// numpy dtype semantics are reproduced deliberately: data is float32;
// scalar factors are applied through double intermediates and cast back
// per element (what numpy's in-place ufuncs with float64 operands do);
// gaussian noise is generated as float64 and cast on store.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <set>
#include <vector>

#include "_rng.h"

namespace py = pybind11;
using seist_rng::RandomState;

namespace {

// float32 pairwise sum, numpy blocking (shared shape with _native.cpp)
float pairwise_sum_f32a(const float* a, ssize_t n) {
  if (n < 8) {
    float s = 0.0f;
    for (ssize_t i = 0; i < n; ++i) s += a[i];
    return s;
  }
  if (n <= 128) {
    float r[8];
    for (int i = 0; i < 8; ++i) r[i] = a[i];
    ssize_t i = 8;
    for (; i + 8 <= n; i += 8) {
      for (int j = 0; j < 8; ++j) r[j] += a[i + j];
    }
    float res =
        ((r[0] + r[1]) + (r[2] + r[3])) + ((r[4] + r[5]) + (r[6] + r[7]));
    for (; i < n; ++i) res += a[i];
    return res;
  }
  ssize_t n2 = n / 2;
  n2 -= n2 % 8;
  return pairwise_sum_f32a(a, n2) + pairwise_sum_f32a(a + n2, n - n2);
}

// np.sum / np.mean iterate in 8192-element reduce buffers combined
// sequentially; pairwise applies within each buffer (verified empirically
// against numpy 2.2 — plain whole-array pairwise diverges for n > 8194)
float npsum_f32(const float* a, ssize_t n) {
  float s = 0.0f;
  for (ssize_t off = 0; off < n; off += 8192) {
    const float cs = pairwise_sum_f32a(a + off, std::min<ssize_t>(8192, n - off));
    s = (off == 0) ? cs : s + cs;
  }
  return s;
}

struct AugParams {
  double min_snr, coda_ratio, p_position_ratio;
  double add_event_rate, add_noise_rate, add_gap_rate, drop_channel_rate;
  double scale_amplitude_rate, pre_emphasis_rate, pre_emphasis_ratio;
  double generate_noise_rate, shift_event_rate;
  long max_event_num, mask_percent, noise_percent;
  long min_event_gap, in_samples, sampling_rate;
  int norm_mode;  // 0 none, 1 max, 2 std
};

using Mat = std::vector<float>;  // (C, L) row-major

struct Event {
  Mat data;
  long C = 0, L = 0;
  std::vector<long> ppks, spks;
  bool cleared = false;  // caller clears the dict's other fields
};

float* row(Event& e, long c) { return e.data.data() + c * e.L; }

// ---- components, draw-order exact with data/preprocess.py ----------------

bool is_noise(const Event& e, const std::vector<double>& snr, double min_snr) {
  const auto& p = e.ppks;
  const auto& s = e.spks;
  bool noise = (p.size() != s.size()) || p.size() < 1 || s.size() < 1;
  if (!noise) {
    long mn = p[0], mx = p[0];
    for (long v : p) { mn = std::min(mn, v); mx = std::max(mx, v); }
    for (long v : s) { mn = std::min(mn, v); mx = std::max(mx, v); }
    bool all_low = true;
    for (double v : snr) all_low &= (v < min_snr);
    noise = mn < 0 || mx >= e.L || all_low;
  }
  for (size_t i = 0; i < std::min(p.size(), s.size()); ++i) {
    noise |= p[i] >= s[i];
  }
  return noise;
}

void pad_phases(std::vector<long>& ppks, std::vector<long>& spks,
                long padding_idx, long num_samples) {
  padding_idx = std::abs(padding_idx);
  std::sort(ppks.begin(), ppks.end());
  std::sort(spks.begin(), spks.end());
  const long np_ = (long)ppks.size(), ns = (long)spks.size();
  long idx = 0;
  while (idx < std::min(np_, ns)) {
    // all(ppk[: idx+1] < spk[-idx-1:]) elementwise
    bool ok = true;
    for (long j = 0; j <= idx; ++j) {
      ok &= ppks[(size_t)j] < spks[(size_t)(ns - idx - 1 + j)];
    }
    if (!ok) break;
    ++idx;
  }
  std::vector<long> p2((size_t)(ns - idx), -padding_idx);
  p2.insert(p2.end(), ppks.begin(), ppks.end());
  ppks = std::move(p2);
  for (long j = 0; j < np_ - idx; ++j) {
    spks.push_back(num_samples + padding_idx);
  }
}

void generate_noise_data(Event& e, RandomState& R, double coda_ratio) {
  if (!e.ppks.empty() && !e.spks.empty()) {
    const size_t n = std::min(e.ppks.size(), e.spks.size());
    for (size_t i = 0; i < n; ++i) {
      const long ppk = e.ppks[i], spk = e.spks[i];
      long coda_end = (long)(spk + coda_ratio * (double)(spk - ppk));
      coda_end = std::max(0L, std::min(coda_end, e.L));
      if (ppk < coda_end) {
        // np.random.randn(C, n) fills row-major
        for (long c = 0; c < e.C; ++c) {
          float* r = row(e, c);
          for (long l = ppk; l < coda_end; ++l) r[l] = (float)R.gauss();
        }
      }
    }
  }
  e.ppks.clear();
  e.spks.clear();
}

void add_event(Event& e, RandomState& R, double coda_ratio, long min_gap) {
  const long ti = R.randint(0, (long)e.ppks.size());
  const long ppk = e.ppks[(size_t)ti], spk = e.spks[(size_t)ti];
  const long coda_end = (long)(spk + coda_ratio * (double)(spk - ppk));
  const long left = coda_end + min_gap;
  const long right = e.L - (spk - ppk) - min_gap;
  if (left < right) {
    const long ppk_add = R.randint(left, right);
    const long spk_add = ppk_add + spk - ppk;
    const long space = std::min(e.L - ppk_add, coda_end - ppk);
    const double scale = R.random_sample();
    for (long c = 0; c < e.C; ++c) {
      float* r = row(e, c);
      for (long l = 0; l < space; ++l) {
        // f32 += f32 * pyfloat: numpy computes in f32 (weak promotion)
        r[ppk_add + l] += r[ppk + l] * (float)scale;
      }
    }
    e.ppks.push_back(ppk_add);
    e.spks.push_back(spk_add);
  }
  std::sort(e.ppks.begin(), e.ppks.end());
  std::sort(e.spks.begin(), e.spks.end());
}

void shift_event(Event& e, RandomState& R) {
  const long shift = R.randint(0, e.L);
  // np.concatenate((data[:, -shift:], data[:, :-shift])): shift==0 keeps
  // the LAST 0 columns... numpy's data[:, -0:] is the WHOLE array, so
  // shift==0 concatenates (all, empty) -> unchanged
  if (shift > 0) {
    Mat out((size_t)(e.C * e.L));
    for (long c = 0; c < e.C; ++c) {
      const float* r = row(e, c);
      float* o = out.data() + c * e.L;
      std::memcpy(o, r + (e.L - shift), (size_t)shift * sizeof(float));
      std::memcpy(o + shift, r, (size_t)(e.L - shift) * sizeof(float));
    }
    e.data = std::move(out);
  }
  for (auto& p : e.ppks) p = ((p + shift) % e.L + e.L) % e.L;
  for (auto& s : e.spks) s = ((s + shift) % e.L + e.L) % e.L;
  std::sort(e.ppks.begin(), e.ppks.end());
  std::sort(e.spks.begin(), e.spks.end());
}

void adjust_amplitude(Event& e) {
  long nonzero = 0;
  for (long c = 0; c < e.C; ++c) {
    const float* r = row(e, c);
    float mx = 0.0f;
    for (long l = 0; l < e.L; ++l) mx = std::max(mx, std::fabs(r[l]));
    if (mx != 0.0f) ++nonzero;
  }
  if (nonzero > 0) {
    // data *= C / count: f64 scalar through numpy in-place -> f32 math
    // with the scalar cast to f32 (weak promotion)
    const float f = (float)((double)e.C / (double)nonzero);
    for (auto& v : e.data) v *= f;
  }
}

void drop_channel(Event& e, RandomState& R) {
  if (e.C < 2) return;
  // np.random.choice(range(1, C)): single draw == randint index
  const long drop_num = 1 + R.randint(0, e.C - 1);
  std::vector<long> candidates((size_t)e.C);
  for (long c = 0; c < e.C; ++c) candidates[(size_t)c] = c;
  for (long k = 0; k < drop_num; ++k) {
    const long idx = R.randint(0, (long)candidates.size());
    const long c = candidates[(size_t)idx];
    candidates.erase(candidates.begin() + idx);
    float* r = row(e, c);
    std::fill(r, r + e.L, 0.0f);
  }
}

void scale_amplitude(Event& e, RandomState& R) {
  if (R.uniform(0.0, 1.0) < 0.5) {
    const float f = (float)R.uniform(1.0, 3.0);
    for (auto& v : e.data) v *= f;
  } else {
    const float f = (float)R.uniform(1.0, 3.0);
    for (auto& v : e.data) v /= f;
  }
}

void pre_emphasis(Event& e, double ratio) {
  for (long c = 0; c < e.C; ++c) {
    float* r = row(e, c);
    // bpf[1:] - ratio*bpf[:-1]: the python-float ratio is weak-promoted,
    // so numpy computes in f32 with ratio cast down; walk backwards so
    // originals are read before overwrite
    const float rf = (float)ratio;
    for (long l = e.L - 1; l >= 1; --l) {
      r[l] = r[l] - rf * r[l - 1];
    }
  }
}

void add_noise(Event& e, RandomState& R) {
  for (long c = 0; c < e.C; ++c) {
    float* r = row(e, c);
    const long snr = R.randint(10, 50);
    // px = np.sum(x**2) / len: f32 pairwise sum of f32 squares, f32 div
    std::vector<float> sq((size_t)e.L);
    for (long l = 0; l < e.L; ++l) sq[(size_t)l] = r[l] * r[l];
    const float px = npsum_f32(sq.data(), e.L) / (float)e.L;
    // pn = px * 10**(-snr/10): snr is a python int, the power a python
    // float -> weak promotion keeps everything float32
    const float pn = px * (float)std::pow(10.0, -(double)snr / 10.0);
    const float amp = std::sqrt(pn);
    // data[c] += randn(L) * amp: f64 noise * f32 scalar -> f64, cast on
    // store
    for (long l = 0; l < e.L; ++l) {
      r[l] = (float)((double)r[l] + R.gauss() * (double)amp);
    }
  }
}

void add_gaps(Event& e, RandomState& R) {
  std::vector<long> phases(e.ppks);
  phases.insert(phases.end(), e.spks.begin(), e.spks.end());
  std::sort(phases.begin(), phases.end());
  long sgt, egt;
  if (!phases.empty()) {
    phases.push_back(e.L - 1);
    std::set<long> uniq(phases.begin(), phases.end());
    phases.assign(uniq.begin(), uniq.end());
    const long ip = R.randint(0, (long)phases.size() - 1);
    sgt = R.randint(phases[(size_t)ip], phases[(size_t)ip + 1]);
    egt = R.randint(sgt, phases[(size_t)ip + 1]);
  } else {
    sgt = R.randint(0, e.L - 1);
    egt = R.randint(sgt + 1, e.L);
  }
  for (long c = 0; c < e.C; ++c) {
    float* r = row(e, c);
    std::fill(r + sgt, r + egt, 0.0f);
  }
}

void add_mask_windows(Event& e, RandomState& R, long percent, long wsize,
                      float mask_value) {
  const long p = std::max(0L, std::min(percent, 100L));
  const long nw = e.L / wsize;
  const long nm = nw * p / 100;
  // np.random.choice(nw, nm, replace=False) == permutation(nw)[:nm] — the
  // permutation draws happen even when nm == 0
  auto sel = R.choice_no_replace(nw, nm);
  for (long i : sel) {
    for (long c = 0; c < e.C; ++c) {
      float* r = row(e, c);
      std::fill(r + i * wsize, r + (i + 1) * wsize, mask_value);
    }
  }
}

void add_noise_windows(Event& e, RandomState& R, long percent, long wsize) {
  const long p = std::max(0L, std::min(percent, 100L));
  const long nw = e.L / wsize;
  const long nb = nw * p / 100;
  auto sel = R.choice_no_replace(nw, nb);
  for (long i : sel) {
    // np.random.randn(C, wsize) row-major
    for (long c = 0; c < e.C; ++c) {
      float* r = row(e, c);
      for (long l = i * wsize; l < (i + 1) * wsize; ++l) {
        r[l] = (float)R.gauss();
      }
    }
  }
}

void data_augmentation(Event& e, RandomState& R, const AugParams& P) {
  if (R.random_sample() < P.generate_noise_rate) {
    generate_noise_data(e, R, P.coda_ratio);
    e.cleared = true;
    if (R.random_sample() < P.drop_channel_rate) {
      drop_channel(e, R);
      adjust_amplitude(e);
    }
    if (R.random_sample() < P.scale_amplitude_rate) scale_amplitude(e, R);
  } else {
    const long iters = P.max_event_num - (long)e.ppks.size();
    for (long i = 0; i < iters; ++i) {
      // python: `np.random.random() < rate and ppks` — draw happens first
      const bool hit = R.random_sample() < P.add_event_rate;
      if (hit && !e.ppks.empty()) {
        add_event(e, R, P.coda_ratio, P.min_event_gap);
      }
    }
    if (R.random_sample() < P.shift_event_rate) shift_event(e, R);
    if (R.random_sample() < P.drop_channel_rate) {
      drop_channel(e, R);
      adjust_amplitude(e);
    }
    if (R.random_sample() < P.scale_amplitude_rate) scale_amplitude(e, R);
    if (R.random_sample() < P.pre_emphasis_rate) {
      pre_emphasis(e, P.pre_emphasis_ratio);
    }
    if (R.random_sample() < P.add_noise_rate) add_noise(e, R);
    if (R.random_sample() < P.add_gap_rate) add_gaps(e, R);
  }
  if (P.mask_percent > 0) {
    add_mask_windows(e, R, P.mask_percent, P.sampling_rate / 2, 1.0f);
  }
  if (P.noise_percent > 0) {
    add_noise_windows(e, R, P.noise_percent, P.sampling_rate / 2);
  }
}

void cut_window(Event& e, RandomState& R, const AugParams& P) {
  const long win = P.in_samples;
  const long input_len = e.L;
  if (0.0 <= P.p_position_ratio && P.p_position_ratio <= 1.0) {
    Mat nd((size_t)(e.C * win), 0.0f);
    long tgt_l = 0, tgt_r = win;
    const long p_idx = e.ppks.at(0);
    long c_l = p_idx - (long)((double)win * P.p_position_ratio);
    long c_r = c_l + win;
    long offset = -c_l;
    if (c_l < 0) {
      tgt_l += -c_l;
      offset += c_l;
      c_l = 0;
    }
    if (c_r > input_len) {
      tgt_r -= c_r - input_len;
      c_r = input_len;
    }
    for (long c = 0; c < e.C; ++c) {
      std::memcpy(nd.data() + c * win + tgt_l, row(e, c) + c_l,
                  (size_t)(c_r - c_l) * sizeof(float));
    }
    offset += tgt_l;
    e.data = std::move(nd);
    e.L = win;
    std::vector<long> p2, s2;
    for (long t : e.ppks) {
      if (0 <= t + offset && t + offset < win) p2.push_back(t + offset);
    }
    for (long t : e.spks) {
      if (0 <= t + offset && t + offset < win) s2.push_back(t + offset);
    }
    e.ppks = std::move(p2);
    e.spks = std::move(s2);
  } else if (input_len > win) {
    long mn = input_len - win;
    for (long t : e.ppks) mn = std::min(mn, t);
    const long hi = std::max(mn - P.min_event_gap, 1L);
    const long c_l = R.randint(0, hi);
    const long c_r = c_l + win;
    Mat nd((size_t)(e.C * win));
    for (long c = 0; c < e.C; ++c) {
      std::memcpy(nd.data() + c * win, row(e, c) + c_l,
                  (size_t)win * sizeof(float));
    }
    e.data = std::move(nd);
    e.L = win;
    std::vector<long> p2, s2;
    for (long t : e.ppks) if (c_l <= t && t < c_r) p2.push_back(t - c_l);
    for (long t : e.spks) if (c_l <= t && t < c_r) s2.push_back(t - c_l);
    e.ppks = std::move(p2);
    e.spks = std::move(s2);
  }
  // input_len < win (float64-promoting pad) is gated off host-side
}

void normalize(Event& e, int mode) {
  for (long c = 0; c < e.C; ++c) {
    float* r = row(e, c);
    const float mean = npsum_f32(r, e.L) / (float)e.L;
    for (long l = 0; l < e.L; ++l) r[l] -= mean;
    if (mode == 1) {
      float mx = r[0];
      for (long l = 1; l < e.L; ++l) mx = std::max(mx, r[l]);
      if (mx == 0.0f) mx = 1.0f;
      for (long l = 0; l < e.L; ++l) r[l] /= mx;
    } else if (mode == 2) {
      const float m2 = npsum_f32(r, e.L) / (float)e.L;
      std::vector<float> sq((size_t)e.L);
      for (long l = 0; l < e.L; ++l) {
        const float d = r[l] - m2;
        sq[(size_t)l] = d * d;
      }
      float sd = std::sqrt(npsum_f32(sq.data(), e.L) / (float)e.L);
      if (sd == 0.0f) sd = 1.0f;
      for (long l = 0; l < e.L; ++l) r[l] /= sd;
    }
  }
}

}  // namespace

// Full DataPreprocessor.process() on one event. Returns
// (data (C, in_samples) float32, ppks, spks, cleared).
py::tuple process_event(py::array_t<float, py::array::c_style> data,
                        std::vector<long> ppks, std::vector<long> spks,
                        std::vector<double> snr, bool augmentation,
                        py::dict params, seist_rng::RandomState* rng) {
  AugParams P;
  P.min_snr = params["min_snr"].cast<double>();
  P.coda_ratio = params["coda_ratio"].cast<double>();
  P.p_position_ratio = params["p_position_ratio"].cast<double>();
  P.add_event_rate = params["add_event_rate"].cast<double>();
  P.add_noise_rate = params["add_noise_rate"].cast<double>();
  P.add_gap_rate = params["add_gap_rate"].cast<double>();
  P.drop_channel_rate = params["drop_channel_rate"].cast<double>();
  P.scale_amplitude_rate = params["scale_amplitude_rate"].cast<double>();
  P.pre_emphasis_rate = params["pre_emphasis_rate"].cast<double>();
  P.pre_emphasis_ratio = params["pre_emphasis_ratio"].cast<double>();
  P.generate_noise_rate = params["generate_noise_rate"].cast<double>();
  P.shift_event_rate = params["shift_event_rate"].cast<double>();
  P.max_event_num = params["max_event_num"].cast<long>();
  P.mask_percent = params["mask_percent"].cast<long>();
  P.noise_percent = params["noise_percent"].cast<long>();
  P.min_event_gap = params["min_event_gap"].cast<long>();
  P.in_samples = params["in_samples"].cast<long>();
  P.sampling_rate = params["sampling_rate"].cast<long>();
  P.norm_mode = params["norm_mode"].cast<int>();

  auto buf = data.unchecked<2>();
  Event e;
  e.C = buf.shape(0);
  e.L = buf.shape(1);
  e.data.resize((size_t)(e.C * e.L));
  std::memcpy(e.data.data(), buf.data(0, 0),
              e.data.size() * sizeof(float));
  e.ppks = std::move(ppks);
  e.spks = std::move(spks);

  if (is_noise(e, snr, P.min_snr)) {
    e.ppks.clear();
    e.spks.clear();
    e.cleared = true;
  }
  pad_phases(e.ppks, e.spks, P.min_event_gap, P.in_samples);
  if (augmentation) data_augmentation(e, *rng, P);
  cut_window(e, *rng, P);
  normalize(e, P.norm_mode);

  py::array_t<float> out({e.C, e.L});
  std::memcpy(out.mutable_unchecked<2>().mutable_data(0, 0), e.data.data(),
              e.data.size() * sizeof(float));
  return py::make_tuple(out, e.ppks, e.spks, e.cleared);
}
