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

// ---------------------------------------------------------------------------
// expand_read: native expand_clip_indent core (pre_lib.py:1128-1239 /
// preprocess/expand.py). One pass over the cigar: gap-expands deletions,
// drops hard clips (and pads), trims soft-clipped columns, indents by the
// reference start, and reverses pw/ip for reverse-strand reads. Returns
// (bases_u32 [view as '<U1' in python], cigar_u8, pw_u8, ip_u8,
//  ccs_idx_i32). The inference worker's hot spot: the python version costs
// ~1.5 ms per 15 kb subread (np.insert copies + per-base cigar expand).
// ---------------------------------------------------------------------------

static constexpr int kCMatch = 0, kCIns = 1, kCDel = 2, kCRefSkip = 3,
                     kCSoft = 4, kCHard = 5, kCPad = 6, kCEq = 7,
                     kCDiff = 8;

static py::tuple expand_read(py::bytes seq_b,
                             py::array_t<int32_t> cigar,  // [n_ops, 2]
                             py::array_t<uint8_t> pw,
                             py::array_t<uint8_t> ip,
                             int64_t pos, bool is_reverse,
                             bool has_tags) {
  char* seq;
  Py_ssize_t n_seq;
  if (PyBytes_AsStringAndSize(seq_b.ptr(), &seq, &n_seq) != 0) {
    throw py::error_already_set();
  }
  auto cg = cigar.unchecked<2>();
  const int64_t n_ops = cg.shape(0);
  auto pwv = pw.unchecked<1>();
  auto ipv = ip.unchecked<1>();
  const int64_t n_tag = pwv.shape(0);

  // Pass 1: column count (excl. hard clips/pads) + soft-clip trim range.
  int64_t total = 0, lead_clip = 0, tail_clip = 0;
  bool seen_body = false;
  for (int64_t i = 0; i < n_ops; ++i) {
    const int op = cg(i, 0);
    const int64_t n = cg(i, 1);
    if (op == kCHard || op == kCPad) continue;
    total += n;
    if (op == kCSoft) {
      if (!seen_body) lead_clip += n;
      else tail_clip += n;
    } else {
      seen_body = true;
      tail_clip = 0;  // only TRAILING clips trim the tail
    }
  }
  const int64_t kept = total - lead_clip - tail_clip;
  const int64_t out_len = pos + kept;

  py::array_t<uint32_t> bases(out_len);
  py::array_t<uint8_t> out_cigar(out_len);
  py::array_t<uint8_t> out_pw(out_len);
  py::array_t<uint8_t> out_ip(out_len);
  py::array_t<int32_t> out_ccs(out_len);
  auto b = static_cast<uint32_t*>(bases.request().ptr);
  auto oc = static_cast<uint8_t*>(out_cigar.request().ptr);
  auto opw = static_cast<uint8_t*>(out_pw.request().ptr);
  auto oip = static_cast<uint8_t*>(out_ip.request().ptr);
  auto occ = static_cast<int32_t*>(out_ccs.request().ptr);

  for (int64_t i = 0; i < pos; ++i) {
    b[i] = ' ';
    oc[i] = kCRefSkip;
    opw[i] = 0;
    oip[i] = 0;
    occ[i] = -1;
  }

  int64_t qpos = 0, rpos = pos, col = 0, out = pos;
  const int64_t lo = lead_clip, hi = lead_clip + kept;
  for (int64_t i = 0; i < n_ops; ++i) {
    const int op = cg(i, 0);
    const int64_t n = cg(i, 1);
    if (op == kCHard || op == kCPad) continue;
    const bool consumes_q =
        (op == kCMatch || op == kCEq || op == kCDiff || op == kCIns ||
         op == kCSoft);
    const bool consumes_r =
        (op == kCMatch || op == kCEq || op == kCDiff || op == kCDel ||
         op == kCRefSkip);
    for (int64_t j = 0; j < n; ++j, ++col) {
      const bool keep = col >= lo && col < hi;
      if (keep) {
        oc[out] = (uint8_t)op;
        if (consumes_q) {
          b[out] = (uint32_t)(unsigned char)seq[qpos];
          if (has_tags) {
            const int64_t t = is_reverse ? (n_tag - 1 - qpos) : qpos;
            const uint8_t pv =
                (t >= 0 && t < n_tag) ? pwv(t) : (uint8_t)0;
            const uint8_t iv =
                (t >= 0 && t < n_tag) ? ipv(t) : (uint8_t)0;
            opw[out] = pv;
            oip[out] = iv;
          } else {
            opw[out] = 0;
            oip[out] = 0;
          }
        } else {
          b[out] = ' ';
          opw[out] = 0;
          oip[out] = 0;
        }
        occ[out] = consumes_r ? (int32_t)rpos : -1;
        ++out;
      }
      if (consumes_q) ++qpos;
      if (consumes_r) ++rpos;
    }
  }
  return py::make_tuple(bases, out_cigar, out_pw, out_ip, out_ccs);
}

PYBIND11_MODULE(_spacing, m) {
  m.doc() = "native gap-spacing state machine + alignment expansion";
  m.def("space_out", &space_out,
        "per-read spaced indices + spaced lengths from insertion masks");
  m.def("expand_read", &expand_read,
        "native expand_clip_indent core: (bases_u32, cigar_u8, pw, ip, "
        "ccs_idx) from (seq, cigar, pw, ip, pos, is_reverse, has_tags)");
}
