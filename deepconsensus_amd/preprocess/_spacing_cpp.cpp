// C++ implementation of the multi-read gap-spacing state machine.
//
// Exact port of the per-column scan semantics in
// deepconsensus_amd/preprocess/read.py (_space_out_python), which mirrors the
// reference pre_lib.py:176-276,1242-1270. This loop is the reference's
// dominant CPU cost during preprocessing; running it natively makes the host
// preprocessing stage keep up with the MI355X device loop.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <vector>

namespace py = pybind11;

namespace {

struct ReadState {
  const uint8_t* ins;
  int64_t len;
  bool is_label;
  int64_t idx_seq = 0;
  int64_t idx_spaced = 0;
  bool done = false;
  int64_t* seq_indices;

  bool oob() const { return idx_seq >= len; }

  bool next_is_insertion() {
    if (is_label) {
      while (!oob() && ins[idx_seq]) {
        seq_indices[idx_seq] = idx_spaced;
        ++idx_seq;
        ++idx_spaced;
      }
      return false;
    }
    if (oob()) return false;
    return ins[idx_seq] != 0;
  }

  void step(bool any_insertions) {
    if (any_insertions && !next_is_insertion()) {
      ++idx_spaced;  // add_gap
    } else {
      if (!oob()) {
        seq_indices[idx_seq] = idx_spaced;
        ++idx_seq;
        ++idx_spaced;
      }
      if (oob()) done = true;
    }
  }
};

}  // namespace

static py::tuple space_out(const std::vector<py::array_t<uint8_t>>& is_ins,
                           py::array_t<uint8_t> is_label) {
  const size_t n = is_ins.size();
  auto lbl = is_label.unchecked<1>();
  std::vector<ReadState> states(n);
  std::vector<py::array_t<int64_t>> outputs;
  outputs.reserve(n);
  for (size_t i = 0; i < n; ++i) {
    auto info = is_ins[i].request();
    const int64_t len = info.shape.empty() ? 0 : info.shape[0];
    outputs.emplace_back(py::array_t<int64_t>(len));
    states[i].ins = static_cast<const uint8_t*>(info.ptr);
    states[i].len = len;
    states[i].is_label = lbl(i) != 0;
    states[i].seq_indices =
        static_cast<int64_t*>(outputs.back().request().ptr);
    if (len == 0) states[i].done = true;
  }

  bool all_done = true;
  for (auto& s : states) all_done &= s.done;
  while (!all_done) {
    bool any_insertions = false;
    for (auto& s : states) {
      if (s.done) continue;
      if (s.next_is_insertion()) {
        any_insertions = true;
        break;
      }
    }
    for (auto& s : states) {
      if (s.done) continue;
      s.step(any_insertions);
    }
    all_done = true;
    for (auto& s : states) all_done &= s.done;
  }

  py::list idx_list;
  py::array_t<int64_t> lens(n);
  auto lens_mut = lens.mutable_unchecked<1>();
  for (size_t i = 0; i < n; ++i) {
    idx_list.append(outputs[i]);
    lens_mut(i) = states[i].idx_spaced;
  }
  return py::make_tuple(idx_list, lens);
}

PYBIND11_MODULE(_spacing, m) {
  m.doc() = "native gap-spacing state machine";
  m.def("space_out", &space_out,
        "per-read spaced indices + spaced lengths from insertion masks");
}
