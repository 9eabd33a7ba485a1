// Batched transition systems for the transition-based parser and NER.
//
// Behavioral contract of spaCy's Cython parser internals (SURVEY.md §2.2 N7:
// upstream spacy/pipeline/_parser_internals/{arc_eager.pyx, ner.pyx,
// _state.pxd}) — re-designed, not translated: one C++ object holds ALL states
// of a batch in struct-of-arrays form, and every API call (features / valid /
// costs / advance) operates on the whole batch so the Python-side per-step
// loop does O(1) native calls per transition step instead of per-state ones.
//
// Arc-eager with the Goldberg & Nivre (2012) dynamic oracle; BILUO NER with
// per-token gold-action costs.  Labeled actions: cost +1 when the arc matches
// gold but the label does not.
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>

#include <algorithm>
#include <cstdint>
#include <vector>

#include "step_iface.h"

#ifdef _OPENMP
#include <omp.h>
#endif

// step_arrays parallelism: bounded so 8 ranks/node don't oversubscribe
static int srx_nthreads() {
#ifdef _OPENMP
  static int n = []() {
    const char* env = getenv("SRX_CPP_THREADS");
    if (env) return std::max(1, atoi(env));
    const char* ws = getenv("WORLD_SIZE");
    int world = ws ? std::max(1, atoi(ws)) : 1;
    return std::min(16, std::max(1, omp_get_max_threads() / (2 * world)));
  }();
  return n;
#else
  return 1;
#endif
}

namespace py = pybind11;

namespace {

constexpr float KInvalid = 1e9f;

// ------------------------------------------------------------------ parser
// Actions: 0=SHIFT, 1=REDUCE, 2..2+L-1 = LEFT-ARC(l), 2+L..2+2L-1 = RIGHT-ARC(l)
struct ParserState {
  std::vector<int32_t> stack;
  int32_t buf = 0;   // index of buffer front
  int32_t len = 0;
  std::vector<int32_t> head;    // -1 = none
  std::vector<int32_t> label;   // -1 = none
  // children bookkeeping for features (two leftmost / two rightmost)
  std::vector<int32_t> l1, l2, r1, r2;

  void init(int32_t n) {
    len = n;
    buf = 0;
    stack.clear();
    head.assign(n, -1);
    label.assign(n, -1);
    l1.assign(n, -1);
    l2.assign(n, -1);
    r1.assign(n, -1);
    r2.assign(n, -1);
  }
  bool final_state() const { return buf >= len && stack.size() <= 1; }
  int32_t s0() const { return stack.empty() ? -1 : stack.back(); }
  int32_t s1() const { return stack.size() < 2 ? -1 : stack[stack.size() - 2]; }
  int32_t s2() const { return stack.size() < 3 ? -1 : stack[stack.size() - 3]; }

  void add_arc(int32_t h, int32_t d, int32_t lab) {
    head[d] = h;
    label[d] = lab;
    if (d < h) {
      if (l1[h] == -1 || d < l1[h]) { l2[h] = l1[h]; l1[h] = d; }
      else if (l2[h] == -1 || d < l2[h]) { l2[h] = d; }
    } else {
      if (r1[h] == -1 || d > r1[h]) { r2[h] = r1[h]; r1[h] = d; }
      else if (r2[h] == -1 || d > r2[h]) { r2[h] = d; }
    }
  }
};

struct ArcEagerBatch : public srx::StepBatchIface {
  int32_t n_labels;
  std::vector<ParserState> states;
  std::vector<int32_t> offsets;               // doc start offset in flat arrays
  std::vector<std::vector<int32_t>> gold_head;   // per doc (empty if no gold)
  std::vector<std::vector<int32_t>> gold_label;
  bool has_gold = false;

  ArcEagerBatch(py::array_t<int32_t, py::array::c_style | py::array::forcecast> lengths,
                int32_t n_labels_, int32_t base_offset = 0)
      : n_labels(n_labels_) {
    if (n_actions() > 256)
      throw std::runtime_error("ArcEagerBatch: > 127 dep labels unsupported");
    auto L = lengths.unchecked<1>();
    int32_t off = base_offset;
    states.resize(L.shape(0));
    offsets.resize(L.shape(0));
    for (py::ssize_t i = 0; i < L.shape(0); i++) {
      states[i].init(L(i));
      offsets[i] = off;
      off += L(i);
    }
  }

  void set_gold(py::array_t<int32_t, py::array::c_style | py::array::forcecast> heads,
                py::array_t<int32_t, py::array::c_style | py::array::forcecast> labels) {
    auto H = heads.unchecked<1>();
    auto Lb = labels.unchecked<1>();
    gold_head.resize(states.size());
    gold_label.resize(states.size());
    for (size_t d = 0; d < states.size(); d++) {
      int32_t off = offsets[d], n = states[d].len;
      gold_head[d].assign(n, -1);
      gold_label[d].assign(n, -1);
      for (int32_t i = 0; i < n; i++) {
        gold_head[d][i] = H(off + i);
        gold_label[d][i] = Lb(off + i);
      }
    }
    has_gold = true;
  }

  int32_t n_actions() const { return 2 + 2 * n_labels; }
  size_t size() const { return states.size(); }

  py::array_t<uint8_t> is_final() const {
    py::array_t<uint8_t> out((py::ssize_t)states.size());
    auto r = out.mutable_unchecked<1>();
    for (size_t i = 0; i < states.size(); i++) r(i) = states[i].final_state() ? 1 : 0;
    return out;
  }

  // 13 context tokens per state, as batch-flat indices (-1 = missing):
  // [S0,S1,S2, B0,B1,B2, L1(S0),L2(S0), R1(S0),R2(S0), L1(S1), R1(S1), head(S0)]
  void fill_features(size_t i, int32_t* out) const {
    const ParserState& st = states[i];
    int32_t off = offsets[i];
    int32_t f[13];
    int32_t s0 = st.s0(), s1 = st.s1(), s2 = st.s2();
    f[0] = s0; f[1] = s1; f[2] = s2;
    f[3] = st.buf < st.len ? st.buf : -1;
    f[4] = st.buf + 1 < st.len ? st.buf + 1 : -1;
    f[5] = st.buf + 2 < st.len ? st.buf + 2 : -1;
    f[6] = s0 >= 0 ? st.l1[s0] : -1;
    f[7] = s0 >= 0 ? st.l2[s0] : -1;
    f[8] = s0 >= 0 ? st.r1[s0] : -1;
    f[9] = s0 >= 0 ? st.r2[s0] : -1;
    f[10] = s1 >= 0 ? st.l1[s1] : -1;
    f[11] = s1 >= 0 ? st.r1[s1] : -1;
    f[12] = s0 >= 0 ? st.head[s0] : -1;
    for (int k = 0; k < 13; k++) out[k] = f[k] >= 0 ? off + f[k] : -1;
  }

  py::array_t<int32_t> features() const {
    py::ssize_t S = (py::ssize_t)states.size();
    py::array_t<int32_t> out({S, (py::ssize_t)13});
    auto r = out.mutable_unchecked<2>();
    for (py::ssize_t i = 0; i < S; i++) fill_features((size_t)i, r.mutable_data(i, 0));
    return out;
  }

  void fill_valid(uint8_t* v, const ParserState& st) const {
    const int32_t A = n_actions();
    bool has_buf = st.buf < st.len;
    bool has_s0 = !st.stack.empty();
    bool s0_has_head = has_s0 && st.head[st.s0()] != -1;
    std::fill(v, v + A, 0);
    if (has_buf) v[0] = 1;                                    // SHIFT
    // REDUCE: s0 has a head; or forced cleanup when the buffer is exhausted
    // (headless pops attach to root), so every non-final state has >=1 valid
    // action and the step loop always terminates.
    if (has_s0 && (s0_has_head || !has_buf)) v[1] = 1;
    bool la_ok = has_s0 && has_buf && !s0_has_head;
    bool ra_ok = has_s0 && has_buf;
    for (int32_t l = 0; l < n_labels; l++) {
      v[2 + l] = la_ok ? 1 : 0;
      v[2 + n_labels + l] = ra_ok ? 1 : 0;
    }
  }

  py::array_t<uint8_t> valid() const {
    py::ssize_t S = (py::ssize_t)states.size(), A = n_actions();
    py::array_t<uint8_t> out({S, A});
    auto r = out.mutable_unchecked<2>();
    for (py::ssize_t i = 0; i < S; i++) fill_valid(r.mutable_data(i, 0), states[i]);
    return out;
  }

  // ---- shared cost computation for one state (Goldberg&Nivre dynamic
  // oracle); writes n_actions() floats, invalid actions get KInvalid.
  void fill_costs(size_t i, uint8_t* v, float* r) const {
    const ParserState& st = states[i];
    const auto& gh = gold_head[i];
    const auto& gl = gold_label[i];
    fill_valid(v, st);
    int32_t b = st.buf < st.len ? st.buf : -1;
    int32_t s0 = st.s0();
    auto in_stack = [&](int32_t t) {
      for (int32_t s : st.stack)
        if (s == t) return true;
      return false;
    };
    float c_shift = 0, c_reduce = 0, c_la = 0, c_ra = 0;
    if (b >= 0) {
      if (gh[b] >= 0 && in_stack(gh[b])) c_shift += 1;
      for (int32_t s : st.stack)
        if (st.head[s] == -1 && gh[s] == b) c_shift += 1;
    }
    if (s0 >= 0) {
      for (int32_t d = st.buf; d < st.len; d++)
        if (gh[d] == s0) c_reduce += 1;
      if (b >= 0) {
        c_la = c_reduce;
        if (gh[s0] >= 0 && gh[s0] > b) c_la += 1;
        if (gh[b] >= 0 && gh[b] != s0 && (in_stack(gh[b]) || gh[b] > b)) c_ra += 1;
        for (int32_t s : st.stack)
          if (st.head[s] == -1 && gh[s] == b) c_ra += 1;
      }
    }
    r[0] = v[0] ? c_shift : KInvalid;
    r[1] = v[1] ? c_reduce : KInvalid;
    for (int32_t l = 0; l < n_labels; l++) {
      float la = c_la, ra = c_ra;
      if (b >= 0 && s0 >= 0) {
        if (gh[s0] == b && gl[s0] != l) la += 1;
        if (gh[b] == s0 && gl[b] != l) ra += 1;
      }
      r[2 + l] = v[2 + l] ? la : KInvalid;
      r[2 + n_labels + l] = v[2 + n_labels + l] ? ra : KInvalid;
    }
  }

  // Goldberg&Nivre dynamic-oracle costs; invalid actions get KInvalid.
  py::array_t<float> costs() const {
    if (!has_gold) throw std::runtime_error("costs() requires set_gold()");
    py::ssize_t S = (py::ssize_t)states.size(), A = n_actions();
    py::array_t<float> out({S, A});
    auto r = out.mutable_unchecked<2>();
    std::vector<uint8_t> v((size_t)A);
    for (py::ssize_t i = 0; i < S; i++) {
      fill_costs((size_t)i, v.data(), r.mutable_data(i, 0));
    }
    return out;
  }

  // ---- fused per-step call: ONE crossing of the pybind boundary returns
  // (active_idx, features, valid, is_gold) for the active states only.
  // is_gold[s,a] = (cost <= min valid cost + eps); empty when !with_gold.
  py::tuple step_arrays(bool with_gold, int64_t pad_row = -1) const {
    std::vector<int32_t> idx;
    idx.reserve(states.size());
    for (size_t i = 0; i < states.size(); i++)
      if (!states[i].final_state()) idx.push_back((int32_t)i);
    py::ssize_t Sa = (py::ssize_t)idx.size();
    const py::ssize_t A = n_actions();
    py::array_t<int32_t> act(Sa);
    py::array_t<int64_t> feats({Sa, (py::ssize_t)13});
    py::array_t<uint8_t> valid_a({Sa, A});
    py::array_t<uint8_t> gold_a({with_gold ? Sa : 0, A});
    std::copy(idx.begin(), idx.end(), act.mutable_data());
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (py::ssize_t k = 0; k < Sa; k++) {
      float crow[256];  // A = 2 + 2*n_labels <= 256 labels supported
      int32_t f32[13];
      size_t i = (size_t)idx[(size_t)k];
      fill_features(i, f32);
      int64_t* fo = feats.mutable_data(k, 0);
      for (int q = 0; q < 13; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a.mutable_data(k, 0);
      if (with_gold) {
        fill_costs(i, v, crow);
        float cmin = KInvalid;
        for (py::ssize_t a = 0; a < A; a++)
          if (v[a] && crow[(size_t)a] < cmin) cmin = crow[(size_t)a];
        uint8_t* g = gold_a.mutable_data(k, 0);
        for (py::ssize_t a = 0; a < A; a++)
          g[a] = (v[a] && crow[(size_t)a] <= cmin + 1e-6f) ? 1 : 0;
      } else {
        fill_valid(v, states[i]);
      }
    }
    return py::make_tuple(act, feats, valid_a, gold_a);
  }

  // ---- StepBatchIface (consumed by the _srx_hip C++ step-loop driver)
  int64_t n_states() const override { return (int64_t)states.size(); }
  int n_feats() const override { return 13; }
  int n_acts() const override { return (int)n_actions(); }
  int64_t max_transitions() const override {
    // buf advances exactly len times (SHIFT/RIGHT-ARC); each push is popped
    // at most once (REDUCE/LEFT-ARC) => <= 2*len transitions per doc.
    int64_t total = 0;
    for (auto& st : states) total += 2 * (int64_t)st.len;
    return total;
  }

  int64_t pack_step(bool with_gold, int64_t pad_row, int32_t* act_idx,
                    int64_t* feats, uint8_t* valid_a, uint8_t* gold_a) override {
    int64_t Sa = 0;
    for (size_t i = 0; i < states.size(); i++)
      if (!states[i].final_state()) act_idx[Sa++] = (int32_t)i;
    const int64_t A = n_actions();
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (int64_t k = 0; k < Sa; k++) {
      float crow[256];
      int32_t f32[13];
      size_t i = (size_t)act_idx[k];
      fill_features(i, f32);
      int64_t* fo = feats + k * 13;
      for (int q = 0; q < 13; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a + k * A;
      if (with_gold) {
        fill_costs(i, v, crow);
        float cmin = KInvalid;
        for (int64_t a = 0; a < A; a++)
          if (v[a] && crow[(size_t)a] < cmin) cmin = crow[(size_t)a];
        uint8_t* g = gold_a + k * A;
        for (int64_t a = 0; a < A; a++)
          g[a] = (v[a] && crow[(size_t)a] <= cmin + 1e-6f) ? 1 : 0;
      } else {
        fill_valid(v, states[i]);
      }
    }
    return Sa;
  }

  inline void apply_action(ParserState& st, int32_t act) {
    if (act == 0) {  // SHIFT
      st.stack.push_back(st.buf);
      st.buf += 1;
    } else if (act == 1) {  // REDUCE
      st.stack.pop_back();
    } else if (act < 2 + n_labels) {  // LEFT-ARC
      int32_t l = act - 2;
      int32_t s0 = st.stack.back();
      st.add_arc(st.buf, s0, l);
      st.stack.pop_back();
    } else {  // RIGHT-ARC
      int32_t l = act - 2 - n_labels;
      int32_t s0 = st.stack.back();
      st.add_arc(s0, st.buf, l);
      st.stack.push_back(st.buf);
      st.buf += 1;
    }
    // Degenerate-state guard: buffer exhausted with stack >1 -> valid()
    // forces REDUCE pops; remaining stack entries keep head -1 (root),
    // like spaCy's unattached tokens defaulting to root.
  }

  void advance_active(const int32_t* act_idx, const int32_t* actions,
                      int64_t n) override {
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (n > 2048)
#endif
    for (int64_t k = 0; k < n; k++) {
      if (actions[k] < 0) continue;
      apply_action(states[(size_t)act_idx[k]], actions[k]);
    }
  }

  // Packed variant: ONE buffer = [feats int64 Sa*13][valid u8 Sa*A]
  // [gold u8 Sa*A (train only)] so the python side does a single H2D copy
  // per transition step (each pageable upload blocks the host ~0.1 ms).
  py::tuple step_arrays_packed(bool with_gold, int64_t pad_row = -1) {
    int64_t Sa0 = 0;
    for (auto& st : states)
      if (!st.final_state()) Sa0++;
    const py::ssize_t A = n_actions();
    const size_t fbytes = (size_t)Sa0 * 13 * 8;
    const size_t vbytes = (size_t)Sa0 * A;
    py::array_t<int32_t> act(Sa0);
    py::array_t<uint8_t> packed((py::ssize_t)(fbytes + vbytes + (with_gold ? vbytes : 0)));
    uint8_t* base = packed.mutable_data();
    pack_step(with_gold, pad_row, act.mutable_data(), (int64_t*)base,
              base + fbytes, base + fbytes + vbytes);
    return py::make_tuple(act, packed, (py::ssize_t)13);
  }

  void advance(py::array_t<int32_t, py::array::c_style | py::array::forcecast> actions) {
    auto a = actions.unchecked<1>();
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (a.shape(0) > 2048)
#endif
    for (py::ssize_t i = 0; i < a.shape(0); i++) {
      ParserState& st = states[i];
      if (st.final_state()) continue;
      int32_t act = a(i);
      if (act < 0) continue;  // explicit no-op (already-final slot)
      apply_action(st, act);
    }
  }

  int64_t handle() { return (int64_t)(intptr_t)static_cast<srx::StepBatchIface*>(this); }

  py::array_t<int32_t> heads() const {
    int32_t total = 0;
    for (auto& st : states) total += st.len;
    py::array_t<int32_t> out((py::ssize_t)total);
    auto r = out.mutable_unchecked<1>();
    int32_t k = 0;
    for (auto& st : states)
      for (int32_t i = 0; i < st.len; i++) r(k++) = st.head[i];
    return out;
  }

  py::array_t<int32_t> labels() const {
    int32_t total = 0;
    for (auto& st : states) total += st.len;
    py::array_t<int32_t> out((py::ssize_t)total);
    auto r = out.mutable_unchecked<1>();
    int32_t k = 0;
    for (auto& st : states)
      for (int32_t i = 0; i < st.len; i++) r(k++) = st.label[i];
    return out;
  }
};

// --------------------------------------------------------------------- NER
// Per-token gold codes: 0 = O; for type t: 1+4t=B, 2+4t=I, 3+4t=L, 4+4t=U.
// Actions: 0 = OUT; for type t: 1+4t=BEGIN, 2+4t=IN, 3+4t=LAST, 4+4t=UNIT.
struct NerState {
  int32_t i = 0;      // current token
  int32_t len = 0;
  int32_t open = -1;  // open entity type or -1
  int32_t open_start = -1;
  std::vector<int32_t> tags;  // emitted per-token action codes

  void init(int32_t n) { i = 0; len = n; open = -1; open_start = -1; tags.assign(n, 0); }
  bool final_state() const { return i >= len; }
};

struct BiluoBatch : public srx::StepBatchIface {
  int32_t n_types;
  std::vector<NerState> states;
  std::vector<int32_t> offsets;
  std::vector<std::vector<int32_t>> gold;  // per-token gold codes
  bool has_gold = false;

  BiluoBatch(py::array_t<int32_t, py::array::c_style | py::array::forcecast> lengths,
             int32_t n_types_, int32_t base_offset = 0)
      : n_types(n_types_) {
    auto L = lengths.unchecked<1>();
    states.resize(L.shape(0));
    offsets.resize(L.shape(0));
    int32_t off = base_offset;
    for (py::ssize_t i = 0; i < L.shape(0); i++) {
      states[i].init(L(i));
      offsets[i] = off;
      off += L(i);
    }
  }

  void set_gold(py::array_t<int32_t, py::array::c_style | py::array::forcecast> codes) {
    auto G = codes.unchecked<1>();
    gold.resize(states.size());
    for (size_t d = 0; d < states.size(); d++) {
      int32_t off = offsets[d], n = states[d].len;
      gold[d].assign(n, 0);
      for (int32_t i = 0; i < n; i++) gold[d][i] = G(off + i);
    }
    has_gold = true;
  }

  int32_t n_actions() const { return 1 + 4 * n_types; }
  size_t size() const { return states.size(); }

  py::array_t<uint8_t> is_final() const {
    py::array_t<uint8_t> out((py::ssize_t)states.size());
    auto r = out.mutable_unchecked<1>();
    for (size_t i = 0; i < states.size(); i++) r(i) = states[i].final_state() ? 1 : 0;
    return out;
  }

  // 6 context tokens: [i-2, i-1, i, i+1, i+2, open_start]  (batch-flat, -1 pad)
  void fill_features(size_t s, int32_t* out) const {
    const NerState& st = states[s];
    int32_t off = offsets[s];
    int32_t f[6] = {st.i - 2, st.i - 1, st.i, st.i + 1, st.i + 2, st.open_start};
    for (int k = 0; k < 6; k++)
      out[k] = (f[k] >= 0 && f[k] < st.len) ? off + f[k] : -1;
  }

  py::array_t<int32_t> features() const {
    py::ssize_t S = (py::ssize_t)states.size();
    py::array_t<int32_t> out({S, (py::ssize_t)6});
    auto r = out.mutable_unchecked<2>();
    for (py::ssize_t s = 0; s < S; s++) fill_features((size_t)s, r.mutable_data(s, 0));
    return out;
  }

  // fused per-step call (same contract as ArcEagerBatch::step_arrays)
  py::tuple step_arrays(bool with_gold, int64_t pad_row = -1) const {
    std::vector<int32_t> idx;
    idx.reserve(states.size());
    for (size_t i = 0; i < states.size(); i++)
      if (!states[i].final_state()) idx.push_back((int32_t)i);
    py::ssize_t Sa = (py::ssize_t)idx.size();
    const py::ssize_t A = n_actions();
    py::array_t<int32_t> act(Sa);
    py::array_t<int64_t> feats({Sa, (py::ssize_t)6});
    py::array_t<uint8_t> valid_a({Sa, A});
    py::array_t<uint8_t> gold_a({with_gold ? Sa : 0, A});
    std::copy(idx.begin(), idx.end(), act.mutable_data());
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (py::ssize_t k = 0; k < Sa; k++) {
      size_t i = (size_t)idx[(size_t)k];
      const NerState& st = states[i];
      int32_t f32[6];
      fill_features(i, f32);
      int64_t* fo = feats.mutable_data(k, 0);
      for (int q = 0; q < 6; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a.mutable_data(k, 0);
      fill_valid(v, st);
      if (with_gold) {
        uint8_t* g = gold_a.mutable_data(k, 0);
        int32_t gcode = st.final_state() ? -2 : gold[i][st.i];
        if (gcode == -1) {
          // MISSING ('-' / unannotated): all-zero gold row -> the loss masks
          // the row out entirely (no positive or negative supervision)
          std::fill(g, g + A, 0);
        } else {
          bool gold_valid = gcode >= 0 && gcode < (int32_t)A && v[gcode];
          for (py::ssize_t a = 0; a < A; a++)
            g[a] = gold_valid ? (a == gcode ? 1 : 0) : v[a];
        }
      }
    }
    return py::make_tuple(act, feats, valid_a, gold_a);
  }

  void fill_valid(uint8_t* v, const NerState& st) const {
    const int32_t A = n_actions();
    std::fill(v, v + A, 0);
    if (st.final_state()) return;
    bool last_tok = st.i == st.len - 1;
    if (st.open < 0) {
      v[0] = 1;  // OUT
      for (int32_t t = 0; t < n_types; t++) {
        if (!last_tok) v[1 + 4 * t] = 1;  // BEGIN needs a following token
        v[4 + 4 * t] = 1;                 // UNIT
      }
    } else {
      if (!last_tok) v[2 + 4 * st.open] = 1;  // IN
      v[3 + 4 * st.open] = 1;                 // LAST
    }
  }

  py::array_t<uint8_t> valid() const {
    py::ssize_t S = (py::ssize_t)states.size(), A = n_actions();
    py::array_t<uint8_t> out({S, A});
    auto r = out.mutable_unchecked<2>();
    for (py::ssize_t i = 0; i < S; i++) fill_valid(r.mutable_data(i, 0), states[i]);
    return out;
  }

  py::array_t<float> costs() const {
    if (!has_gold) throw std::runtime_error("costs() requires set_gold()");
    py::ssize_t S = (py::ssize_t)states.size(), A = n_actions();
    py::array_t<float> out({S, A});
    auto r = out.mutable_unchecked<2>();
    std::vector<uint8_t> v((size_t)A);
    for (py::ssize_t s = 0; s < S; s++) {
      const NerState& st = states[s];
      fill_valid(v.data(), st);
      for (py::ssize_t a = 0; a < A; a++) {
        if (!v[a]) { r(s, a) = KInvalid; continue; }
        int32_t g = st.final_state() ? -2 : gold[s][st.i];
        // g == -1 (missing annotation): every valid action is free
        r(s, a) = (g == -1 || (int32_t)a == g) ? 0.0f : 1.0f;
      }
    }
    return out;
  }

  // ---- StepBatchIface
  int64_t n_states() const override { return (int64_t)states.size(); }
  int n_feats() const override { return 6; }
  int n_acts() const override { return (int)n_actions(); }
  int64_t max_transitions() const override {
    int64_t total = 0;
    for (auto& st : states) total += (int64_t)st.len;  // one action per token
    return total;
  }

  int64_t pack_step(bool with_gold, int64_t pad_row, int32_t* act_idx,
                    int64_t* feats, uint8_t* valid_a, uint8_t* gold_a) override {
    int64_t Sa = 0;
    for (size_t i = 0; i < states.size(); i++)
      if (!states[i].final_state()) act_idx[Sa++] = (int32_t)i;
    const int64_t A = n_actions();
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (int64_t k = 0; k < Sa; k++) {
      size_t i = (size_t)act_idx[k];
      const NerState& st = states[i];
      int32_t f32[6];
      fill_features(i, f32);
      int64_t* fo = feats + k * 6;
      for (int q = 0; q < 6; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a + k * A;
      fill_valid(v, st);
      if (with_gold) {
        uint8_t* g = gold_a + k * A;
        int32_t gcode = st.final_state() ? -2 : gold[i][st.i];
        if (gcode == -1) {
          std::fill(g, g + A, 0);  // missing: row excluded from the loss
        } else {
          bool gold_valid = gcode >= 0 && gcode < (int32_t)A && v[gcode];
          for (int64_t a = 0; a < A; a++)
            g[a] = gold_valid ? (a == gcode ? 1 : 0) : v[a];
        }
      }
    }
    return Sa;
  }

  inline void apply_action(NerState& st, int32_t act) {
    st.tags[st.i] = act;
    if (act == 0) {
      st.open = -1; st.open_start = -1;
    } else {
      int32_t t = (act - 1) / 4;
      int32_t kind = (act - 1) % 4;  // 0=B,1=I,2=L,3=U
      if (kind == 0) { st.open = t; st.open_start = st.i; }
      else if (kind == 1) { /* stays open */ }
      else { st.open = -1; st.open_start = -1; }
    }
    st.i += 1;
  }

  void advance_active(const int32_t* act_idx, const int32_t* actions,
                      int64_t n) override {
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (n > 2048)
#endif
    for (int64_t k = 0; k < n; k++) {
      if (actions[k] < 0) continue;
      apply_action(states[(size_t)act_idx[k]], actions[k]);
    }
  }

  py::tuple step_arrays_packed(bool with_gold, int64_t pad_row = -1) {
    int64_t Sa0 = 0;
    for (auto& st : states)
      if (!st.final_state()) Sa0++;
    const py::ssize_t A = n_actions();
    const size_t fbytes = (size_t)Sa0 * 6 * 8;
    const size_t vbytes = (size_t)Sa0 * A;
    py::array_t<int32_t> act(Sa0);
    py::array_t<uint8_t> packed((py::ssize_t)(fbytes + vbytes + (with_gold ? vbytes : 0)));
    uint8_t* base = packed.mutable_data();
    pack_step(with_gold, pad_row, act.mutable_data(), (int64_t*)base,
              base + fbytes, base + fbytes + vbytes);
    return py::make_tuple(act, packed, (py::ssize_t)6);
  }

  void advance(py::array_t<int32_t, py::array::c_style | py::array::forcecast> actions) {
    auto a = actions.unchecked<1>();
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (a.shape(0) > 2048)
#endif
    for (py::ssize_t s = 0; s < a.shape(0); s++) {
      NerState& st = states[s];
      if (st.final_state()) continue;
      int32_t act = a(s);
      if (act < 0) continue;
      apply_action(st, act);
    }
  }

  int64_t handle() { return (int64_t)(intptr_t)static_cast<srx::StepBatchIface*>(this); }

  py::array_t<int32_t> tags() const {
    int32_t total = 0;
    for (auto& st : states) total += st.len;
    py::array_t<int32_t> out((py::ssize_t)total);
    auto r = out.mutable_unchecked<1>();
    int32_t k = 0;
    for (auto& st : states)
      for (int32_t i = 0; i < st.len; i++) r(k++) = st.tags[i];
    return out;
  }
};

}  // namespace

void init_transitions(py::module_& m) {
  py::class_<ArcEagerBatch>(m, "ArcEagerBatch")
      .def(py::init<py::array_t<int32_t, py::array::c_style | py::array::forcecast>, int32_t, int32_t>(),
           py::arg("lengths"), py::arg("n_labels"), py::arg("base_offset") = 0)
      .def("set_gold", &ArcEagerBatch::set_gold, py::arg("heads"), py::arg("labels"))
      .def_property_readonly("n_actions", &ArcEagerBatch::n_actions)
      .def("__len__", &ArcEagerBatch::size)
      .def("is_final", &ArcEagerBatch::is_final)
      .def("features", &ArcEagerBatch::features)
      .def("valid", &ArcEagerBatch::valid)
      .def("costs", &ArcEagerBatch::costs)
      .def("advance", &ArcEagerBatch::advance)
      .def("step_arrays", &ArcEagerBatch::step_arrays, py::arg("with_gold"), py::arg("pad_row") = -1)
      .def("step_arrays_packed", &ArcEagerBatch::step_arrays_packed, py::arg("with_gold"), py::arg("pad_row") = -1)
      .def("handle", &ArcEagerBatch::handle)
      .def("heads", &ArcEagerBatch::heads)
      .def("labels", &ArcEagerBatch::labels);

  py::class_<BiluoBatch>(m, "BiluoBatch")
      .def(py::init<py::array_t<int32_t, py::array::c_style | py::array::forcecast>, int32_t, int32_t>(),
           py::arg("lengths"), py::arg("n_types"), py::arg("base_offset") = 0)
      .def("set_gold", &BiluoBatch::set_gold, py::arg("codes"))
      .def_property_readonly("n_actions", &BiluoBatch::n_actions)
      .def("__len__", &BiluoBatch::size)
      .def("is_final", &BiluoBatch::is_final)
      .def("features", &BiluoBatch::features)
      .def("valid", &BiluoBatch::valid)
      .def("costs", &BiluoBatch::costs)
      .def("advance", &BiluoBatch::advance)
      .def("step_arrays", &BiluoBatch::step_arrays, py::arg("with_gold"), py::arg("pad_row") = -1)
      .def("step_arrays_packed", &BiluoBatch::step_arrays_packed, py::arg("with_gold"), py::arg("pad_row") = -1)
      .def("handle", &BiluoBatch::handle)
      .def("tags", &BiluoBatch::tags);
}
