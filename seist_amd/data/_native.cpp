// K20: native data-pipeline workers.
//
// The loader-side hot loops of the preprocessor (reference
// training/preprocess.py:224-242 normalize, :544-683 soft labels) in C++,
// called from seist_amd/data/preprocess.py inside the DataLoader worker
// processes. Summation uses numpy's pairwise algorithm (8-way unrolled
// base case, 128-element blocks) so the float32 demean/std results match
// the numpy path bit-for-bit; the label-window shape itself is computed
// once in numpy and passed in, so no libm difference can leak in.

#include <pybind11/numpy.h>
#include <pybind11/stl.h>
#include <pybind11/pybind11.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <vector>

#include "_rng.h"

namespace py = pybind11;

namespace {

// numpy's pairwise summation (numpy/core/src/umath/loops_utils.h.src
// structure): base case n <= 8 unrolled; n <= 128 accumulated in 8 lanes;
// larger n split in half (lower half rounded to a multiple of 8).
template <typename T>
double pairwise_sum(const T* a, ssize_t n) {
  if (n < 8) {
    double s = 0.0;
    for (ssize_t i = 0; i < n; ++i) s += (double)a[i];
    return s;
  }
  if (n <= 128) {
    double r[8];
    for (int i = 0; i < 8; ++i) r[i] = (double)a[i];
    ssize_t i = 8;
    for (; i + 8 <= n; i += 8) {
      for (int j = 0; j < 8; ++j) r[j] += (double)a[i + j];
    }
    double res =
        ((r[0] + r[1]) + (r[2] + r[3])) + ((r[4] + r[5]) + (r[6] + r[7]));
    for (; i < n; ++i) res += (double)a[i];  // remainder after the tree
    return res;
  }
  ssize_t n2 = n / 2;
  n2 -= n2 % 8;
  return pairwise_sum(a, n2) + pairwise_sum(a + n2, n - n2);
}

// float32 pairwise sum in float32 precision (numpy sums float32 arrays in
// float32 with pairwise blocking — the accumulator dtype is the array
// dtype, not double)
float pairwise_sum_f32(const float* a, ssize_t n) {
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
    for (; i < n; ++i) res += a[i];  // remainder after the tree
    return res;
  }
  ssize_t n2 = n / 2;
  n2 -= n2 % 8;
  return pairwise_sum_f32(a, n2) + pairwise_sum_f32(a + n2, n - n2);
}

// np.sum / np.mean iterate in 8192-element reduce buffers combined
// sequentially; pairwise applies within each buffer (verified empirically
// against numpy 2.2 — plain whole-array pairwise diverges for n > 8194)
float npsum_f32(const float* a, ssize_t n) {
  float s = 0.0f;
  for (ssize_t off = 0; off < n; off += 8192) {
    const float cs = pairwise_sum_f32(a + off, std::min<ssize_t>(8192, n - off));
    s = (off == 0) ? cs : s + cs;
  }
  return s;
}

// demean + optional per-channel scaling, matching
// DataPreprocessor._normalize (numpy semantics: mean/std computed in the
// array dtype via pairwise summation; np.std is the biased estimator)
void normalize(py::array_t<float, py::array::c_style> data, int mode) {
  auto buf = data.mutable_unchecked<2>();
  const ssize_t C = buf.shape(0), L = buf.shape(1);
  for (ssize_t c = 0; c < C; ++c) {
    float* row = buf.mutable_data(c, 0);
    const float mean = npsum_f32(row, L) / (float)L;
    for (ssize_t l = 0; l < L; ++l) row[l] -= mean;
    if (mode == 1) {  // "max"
      float mx = row[0];
      for (ssize_t l = 1; l < L; ++l) mx = std::max(mx, row[l]);
      if (mx == 0.0f) mx = 1.0f;
      for (ssize_t l = 0; l < L; ++l) row[l] /= mx;
    } else if (mode == 2) {  // "std"
      // np.std: mean of squared deviations (row is already demeaned, but
      // numpy recomputes the mean of the demeaned row — reproduce that)
      const float m2 = npsum_f32(row, L) / (float)L;
      std::vector<float> sq((size_t)L);
      for (ssize_t l = 0; l < L; ++l) {
        const float d = row[l] - m2;
        sq[(size_t)l] = d * d;
      }
      float sd = std::sqrt(npsum_f32(sq.data(), L) / (float)L);
      if (sd == 0.0f) sd = 1.0f;
      for (ssize_t l = 0; l < L; ++l) row[l] /= sd;
    }
  }
}

// DataPreprocessor._rasterize: sum the (precomputed) label window at each
// index with edge clipping; float64 like the numpy path
py::array_t<double> rasterize(const std::vector<long>& idxs, long length,
                              long width,
                              py::array_t<double, py::array::c_style> window) {
  auto win = window.unchecked<1>();
  py::array_t<double> out((ssize_t)length);
  auto ob = out.mutable_unchecked<1>();
  for (long i = 0; i < length; ++i) ob(i) = 0.0;
  const long left = width / 2;
  const long right = width - left;
  for (const long idx : idxs) {
    if (idx < 0) continue;
    if (idx - left < 0) {
      const long n = idx + right + 1;
      const long off = width + 1 - n;
      for (long i = 0; i < n; ++i) ob(i) += win(off + i);
    } else if (idx + right <= length - 1) {
      for (long i = 0; i < width + 1; ++i) ob(idx - left + i) += win(i);
    } else if (idx <= length - 1) {
      const long n = length - (idx - left);
      for (long i = 0; i < n; ++i) ob(length - n + i) += win(i);
    }
  }
  return out;
}

// d<channel> label: out[0] = 0, out[1:] = np.diff(ch)
py::array_t<float> diff_label(py::array_t<float, py::array::c_style> ch) {
  auto x = ch.unchecked<1>();
  const ssize_t L = x.shape(0);
  py::array_t<float> out(L);
  auto ob = out.mutable_unchecked<1>();
  ob(0) = 0.0f;
  for (ssize_t i = 1; i < L; ++i) ob(i) = x(i) - x(i - 1);
  return out;
}

// cal_snr for the SOS reader (utils/misc.py parity): 10*log10(Ps/Pn) over
// the windows around the pick
double cal_snr_native(py::array_t<float, py::array::c_style> data, long pat,
                      long window) {
  auto x = data.unchecked<1>();
  const ssize_t L = x.shape(0);
  const long s0 = std::max(0L, pat - window);
  const long s1 = std::min((long)L, pat + window);
  if (pat <= s0 || s1 <= pat) return 0.0;
  double pn = 0.0, ps = 0.0;
  for (long i = s0; i < pat; ++i) pn += (double)x(i) * x(i);
  for (long i = pat; i < s1; ++i) ps += (double)x(i) * x(i);
  pn /= (double)(pat - s0);
  ps /= (double)(s1 - pat);
  if (pn <= 0.0 || ps <= 0.0) return 0.0;
  return 10.0 * std::log10(ps / pn);
}

}  // namespace

// full per-sample pipeline (implemented in _augment.cpp)
py::tuple process_event(py::array_t<float, py::array::c_style> data,
                        std::vector<long> ppks, std::vector<long> spks,
                        std::vector<double> snr, bool augmentation,
                        py::dict params, seist_rng::RandomState* rng);

// Python-visible wrapper for tests / the C++ augmentation chain
class PyRandomState {
 public:
  explicit PyRandomState(uint32_t seed) : rs_(seed) {}
  double random_sample() { return rs_.random_sample(); }
  double uniform(double lo, double hi) { return rs_.uniform(lo, hi); }
  long randint(long lo, long hi) { return rs_.randint(lo, hi); }
  double gauss() { return rs_.gauss(); }
  py::array_t<double> standard_normal(long n) {
    py::array_t<double> out(n);
    auto b = out.mutable_unchecked<1>();
    for (long i = 0; i < n; ++i) b(i) = rs_.gauss();
    return out;
  }
  std::vector<long> permutation(long n) { return rs_.permutation(n); }
  std::vector<long> choice_no_replace(long n, long size) {
    return rs_.choice_no_replace(n, size);
  }
  void set_state(const std::vector<uint32_t>& keys, int pos,
                 bool has_gauss, double gauss) {
    rs_.set_state(keys, pos, has_gauss, gauss);
  }
  py::tuple get_state() const {
    std::vector<uint32_t> keys;
    int pos;
    bool hg;
    double g;
    rs_.get_state(&keys, &pos, &hg, &g);
    return py::make_tuple(keys, pos, hg, g);
  }
  seist_rng::RandomState* raw() { return &rs_; }

 private:
  seist_rng::RandomState rs_;
};

PYBIND11_MODULE(_native_data, m) {
  m.doc() = "seist_amd native data-pipeline workers (K20)";
  py::class_<PyRandomState>(m, "RandomState")
      .def(py::init<uint32_t>())
      .def("random_sample", &PyRandomState::random_sample)
      .def("uniform", &PyRandomState::uniform)
      .def("randint", &PyRandomState::randint)
      .def("gauss", &PyRandomState::gauss)
      .def("standard_normal", &PyRandomState::standard_normal)
      .def("permutation", &PyRandomState::permutation)
      .def("choice_no_replace", &PyRandomState::choice_no_replace)
      .def("set_state", &PyRandomState::set_state)
      .def("get_state", &PyRandomState::get_state);
  m.def("process_event",
        [](py::array_t<float, py::array::c_style> data,
           std::vector<long> ppks, std::vector<long> spks,
           std::vector<double> snr, bool augmentation, py::dict params,
           PyRandomState& rng) {
          return process_event(data, std::move(ppks), std::move(spks),
                               std::move(snr), augmentation, params,
                               rng.raw());
        },
        "full per-sample preprocessing pipeline (noise gate, pad, "
        "augment, cut, normalize) — bit-exact with the numpy path");
  m.def("normalize", &normalize, "in-place demean + max/std normalize",
        py::arg("data"), py::arg("mode"));
  m.def("rasterize", &rasterize, "sum label window at indices",
        py::arg("idxs"), py::arg("length"), py::arg("width"),
        py::arg("window"));
  m.def("diff_label", &diff_label, "first-difference channel label");
  m.def("cal_snr", &cal_snr_native, "pick-window SNR in dB");
}
