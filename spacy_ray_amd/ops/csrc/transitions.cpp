// Batched transition systems for the transition-based parser and NER.
//
// Behavioral contract of spaCy's Cython parser internals (SURVEY.md §2.2 N7:
// upstream spacy/pipeline/_parser_internals/{arc_eager.pyx, ner.pyx,
// _state.pxd}) — re-designed, not translated: one C++ object holds ALL states
// of a batch in STRUCT-OF-ARRAYS form (flat token-indexed arrays with doc
// offsets — a 50k-doc batch is ~12 allocations, not 400k std::vectors; batch
// construction measured ~50 ms/step at 1M words in the AoS round-1 layout),
// and every API call (features / valid / costs / advance) operates on the
// whole batch so the per-step loop does O(1) native calls per transition
// step instead of per-state ones.  The srx::StepBatchIface seam lets the
// _srx_hip C++ step-loop driver run the whole loop without Python.
//
// Arc-eager with the Goldberg & Nivre (2012) dynamic oracle; BILUO NER with
// per-token gold-action costs.  Labeled actions: cost +1 when the arc matches
// gold but the label does not.
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
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
    // measured on the 256-core MI355X box: 16 threads beat 32/64 by ~35%
    // end-to-end (fork/join + NUMA costs dominate past 16)
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
// SoA layout: every token-indexed array is FLAT over the batch; doc d's
// tokens live at [off[d], off[d]+len[d]).  The per-doc stack is a flat
// arena slice of the same extent (stack depth <= len).  Stored head/label/
// child indices are DOC-LOCAL (like the round-1 AoS layout) so the oracle
// math is unchanged; fill_features adds the doc offset for the GPU gather.
//
// Actions: 0=SHIFT, 1=REDUCE, 2..2+L-1=LEFT-ARC(l), 2+L..2+2L-1=RIGHT-ARC(l),
// [2+2L=BREAK when use_break].  BREAK (sentence boundary, spaCy USE_BREAK
// contract re-designed — see docs/PARITY.md): valid when at least one token
// is consumed, the buffer is non-empty and B0 is not already marked; it
// marks B0 as a sentence start and puts the state into CLEANUP mode — only
// REDUCE is valid until the stack empties (headless pops stay attached to
// root, exactly like end-of-buffer cleanup), so no arc can cross the
// boundary.  Oracle: BREAK costs (gold sent_start[B0] ? 0 : 1) + the number
// of gold arcs that would cross the boundary (stack tokens with gold heads
// or gold children at/after B0); SHIFT and RIGHT-ARC over an unmarked gold
// boundary cost +1 (they pull B0 into the current sentence).
struct ArcEagerBatch : public srx::StepBatchIface {
  int32_t n_labels;
  bool use_break = false;
  int64_t n_docs = 0, total = 0;
  int32_t base_offset = 0;
  std::vector<int32_t> off;     // [n_docs + 1]
  std::vector<int32_t> len;     // [n_docs]
  std::vector<int32_t> buf;     // [n_docs] buffer front (doc-local)
  std::vector<int32_t> ssize;   // [n_docs] stack size
  std::vector<int32_t> stack;   // flat arena [total]
  std::vector<int32_t> head, label, l1, l2, r1, r2;  // flat [total], doc-local ids
  std::vector<int32_t> gold_head, gold_label;        // flat [total]
  std::vector<int32_t> gold_sent;                    // flat [total]; empty = none
  std::vector<int32_t> sent_out;                     // flat [total] predicted starts
  std::vector<uint8_t> pending_break;                // [n_docs] cleanup mode
  bool has_gold = false;
  // O(1)-oracle bookkeeping (fill_costs was 28 ms/step at 1M words with the
  // O(stack + buffer) formulation):
  std::vector<uint8_t> on_stack;        // flat [total]
  std::vector<int32_t> gold_kids_buf;   // [total] # gold children still in buffer
  std::vector<int32_t> kids_off;        // CSR [total+1]: gold children lists
  std::vector<int32_t> kids;            // CSR payload (doc-local child ids)

  ArcEagerBatch(py::array_t<int32_t, py::array::c_style | py::array::forcecast> lengths,
                int32_t n_labels_, int32_t base_offset_ = 0,
                bool use_break_ = false)
      : n_labels(n_labels_), use_break(use_break_), base_offset(base_offset_) {
    if (n_actions() > 256)
      throw std::runtime_error(
          "ArcEagerBatch: action space > 256 (more than ~127 dep labels); "
          "the per-state cost buffer is fixed at 256 actions — split the "
          "label set or raise the cap in transitions.cpp/pack_step");
    auto L = lengths.unchecked<1>();
    n_docs = L.shape(0);
    off.resize(n_docs + 1);
    len.resize(n_docs);
    off[0] = 0;
    for (int64_t i = 0; i < n_docs; i++) {
      len[i] = L(i);
      off[i + 1] = off[i] + L(i);
    }
    total = off[n_docs];
    buf.assign(n_docs, 0);
    ssize.assign(n_docs, 0);
    stack.resize(total);
    head.assign(total, -1);
    label.assign(total, -1);
    l1.assign(total, -1);
    l2.assign(total, -1);
    r1.assign(total, -1);
    r2.assign(total, -1);
    on_stack.assign(total, 0);
    sent_out.assign(total, 0);
    pending_break.assign(n_docs, 0);
  }

  void set_sent_gold(py::array_t<int32_t, py::array::c_style | py::array::forcecast> sents) {
    auto S = sents.unchecked<1>();
    if ((int64_t)S.shape(0) < base_offset + total)
      throw std::runtime_error("set_sent_gold: array shorter than batch");
    gold_sent.resize(total);
    std::memcpy(gold_sent.data(), S.data(0) + base_offset, total * sizeof(int32_t));
  }

  void set_gold(py::array_t<int32_t, py::array::c_style | py::array::forcecast> heads,
                py::array_t<int32_t, py::array::c_style | py::array::forcecast> labels) {
    // gold arrays are GLOBAL (whole batch, all shards); this shard's slice
    // starts at base_offset (pipes.py passes the full concatenated gold)
    auto H = heads.unchecked<1>();
    auto Lb = labels.unchecked<1>();
    if ((int64_t)H.shape(0) < base_offset + total)
      throw std::runtime_error("set_gold: heads shorter than batch");
    gold_head.resize(total);
    gold_label.resize(total);
    std::memcpy(gold_head.data(), H.data(0) + base_offset, total * sizeof(int32_t));
    std::memcpy(gold_label.data(), Lb.data(0) + base_offset, total * sizeof(int32_t));
    has_gold = true;
    // gold-children CSR + in-buffer counters (tokens start in the buffer):
    // per-doc-independent count/fill passes run under OpenMP; only the 1M-add
    // prefix sum is serial.
    gold_kids_buf.assign(total, 0);
    kids_off.resize(total + 1);
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (n_docs > 2048)
#endif
    for (int64_t d = 0; d < n_docs; d++) {
      const int64_t o = off[d];
      const int32_t n = len[d];
      for (int32_t i = 0; i < n; i++) {
        int32_t h = gold_head[o + i];
        if (h >= 0 && h < n) gold_kids_buf[o + h]++;
      }
    }
    kids_off[0] = 0;
    for (int64_t t = 0; t < total; t++)
      kids_off[t + 1] = kids_off[t] + gold_kids_buf[t];
    kids.resize(kids_off[total]);
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (n_docs > 2048)
#endif
    for (int64_t d = 0; d < n_docs; d++) {
      const int64_t o = off[d];
      const int32_t n = len[d];
      std::vector<int32_t> cur(kids_off.begin() + o, kids_off.begin() + o + n);
      for (int32_t i = 0; i < n; i++) {
        int32_t h = gold_head[o + i];
        if (h >= 0 && h < n) kids[cur[h]++] = i;
      }
    }
  }

  int32_t n_actions() const { return 2 + 2 * n_labels + (use_break ? 1 : 0); }
  size_t size() const { return (size_t)n_docs; }

  inline bool final_state(int64_t d) const {
    return buf[d] >= len[d] && ssize[d] <= 1;
  }
  inline int32_t s0(int64_t d) const {
    return ssize[d] ? stack[off[d] + ssize[d] - 1] : -1;
  }
  inline int32_t s1(int64_t d) const {
    return ssize[d] >= 2 ? stack[off[d] + ssize[d] - 2] : -1;
  }
  inline int32_t s2(int64_t d) const {
    return ssize[d] >= 3 ? stack[off[d] + ssize[d] - 3] : -1;
  }

  inline void add_arc(int64_t d, int32_t h, int32_t dep, int32_t lab) {
    const int64_t o = off[d];
    head[o + dep] = h;
    label[o + dep] = lab;
    if (dep < h) {
      if (l1[o + h] == -1 || dep < l1[o + h]) { l2[o + h] = l1[o + h]; l1[o + h] = dep; }
      else if (l2[o + h] == -1 || dep < l2[o + h]) { l2[o + h] = dep; }
    } else {
      if (r1[o + h] == -1 || dep > r1[o + h]) { r2[o + h] = r1[o + h]; r1[o + h] = dep; }
      else if (r2[o + h] == -1 || dep > r2[o + h]) { r2[o + h] = dep; }
    }
  }

  py::array_t<uint8_t> is_final() const {
    py::array_t<uint8_t> out((py::ssize_t)n_docs);
    auto r = out.mutable_unchecked<1>();
    for (int64_t i = 0; i < n_docs; i++) r(i) = final_state(i) ? 1 : 0;
    return out;
  }

  // 13 context tokens per state, as batch-flat indices (-1 = missing):
  // [S0,S1,S2, B0,B1,B2, L1(S0),L2(S0), R1(S0),R2(S0), L1(S1), R1(S1), head(S0)]
  void fill_features(int64_t d, int32_t* out) const {
    const int64_t o = off[d];
    int32_t f[13];
    int32_t v0 = s0(d), v1 = s1(d), v2 = s2(d);
    f[0] = v0; f[1] = v1; f[2] = v2;
    f[3] = buf[d] < len[d] ? buf[d] : -1;
    f[4] = buf[d] + 1 < len[d] ? buf[d] + 1 : -1;
    f[5] = buf[d] + 2 < len[d] ? buf[d] + 2 : -1;
    f[6] = v0 >= 0 ? l1[o + v0] : -1;
    f[7] = v0 >= 0 ? l2[o + v0] : -1;
    f[8] = v0 >= 0 ? r1[o + v0] : -1;
    f[9] = v0 >= 0 ? r2[o + v0] : -1;
    f[10] = v1 >= 0 ? l1[o + v1] : -1;
    f[11] = v1 >= 0 ? r1[o + v1] : -1;
    f[12] = v0 >= 0 ? head[o + v0] : -1;
    const int32_t ob = base_offset + (int32_t)o;
    for (int k = 0; k < 13; k++) out[k] = f[k] >= 0 ? ob + f[k] : -1;
  }

  py::array_t<int32_t> features() const {
    py::array_t<int32_t> out({(py::ssize_t)n_docs, (py::ssize_t)13});
    auto r = out.mutable_unchecked<2>();
    for (int64_t i = 0; i < n_docs; i++) fill_features(i, r.mutable_data(i, 0));
    return out;
  }

  void fill_valid(uint8_t* v, int64_t d) const {
    const int32_t A = n_actions();
    bool has_buf = buf[d] < len[d];
    bool has_s0 = ssize[d] > 0;
    bool s0_has_head = has_s0 && head[off[d] + s0(d)] != -1;
    std::fill(v, v + A, 0);
    if (use_break && pending_break[d] && has_s0) {
      v[1] = 1;  // cleanup mode after BREAK: force pops until stack empties
      return;
    }
    if (has_buf) v[0] = 1;                                    // SHIFT
    // REDUCE: s0 has a head; or forced cleanup when the buffer is exhausted
    // (headless pops attach to root), so every non-final state has >=1 valid
    // action and the step loop always terminates.
    if (has_s0 && (s0_has_head || !has_buf)) v[1] = 1;
    bool la_ok = has_s0 && has_buf && !s0_has_head;
    bool ra_ok = has_s0 && has_buf;
    if (la_ok) std::memset(v + 2, 1, (size_t)n_labels);
    if (ra_ok) std::memset(v + 2 + n_labels, 1, (size_t)n_labels);
    if (use_break)
      v[2 + 2 * n_labels] =
          (has_buf && buf[d] > 0 && !sent_out[off[d] + buf[d]]) ? 1 : 0;
  }

  py::array_t<uint8_t> valid() const {
    py::ssize_t A = n_actions();
    py::array_t<uint8_t> out({(py::ssize_t)n_docs, A});
    auto r = out.mutable_unchecked<2>();
    for (int64_t i = 0; i < n_docs; i++) fill_valid(r.mutable_data(i, 0), i);
    return out;
  }

  // ---- shared cost computation for one state (Goldberg&Nivre dynamic
  // oracle); writes n_actions() floats, invalid actions get KInvalid.
  // O(1) + O(deg(b)) per state via the on_stack bitmap, the per-head
  // gold-children-in-buffer counters and the gold-children CSR (replaces
  // the O(stack + buffer) scans — 28 ms/step of pack time at 1M words).
  void fill_costs(int64_t d, uint8_t* v, float* r) const {
    const int64_t o = off[d];
    const int32_t* gh = gold_head.data() + o;
    const int32_t* gl = gold_label.data() + o;
    fill_valid(v, d);
    int32_t b = buf[d] < len[d] ? buf[d] : -1;
    int32_t v0 = s0(d);
    float c_shift = 0, c_reduce = 0, c_la = 0, c_ra = 0;
    if (b >= 0) {
      // gold head of b sits on the stack: SHIFT/RIGHT-ARC can lose the arc
      bool ghb_on_stack =
          gh[b] >= 0 && gh[b] < len[d] && on_stack[o + gh[b]];
      if (ghb_on_stack) c_shift += 1;
      // headless stack tokens whose gold head is b (gold children of b on
      // the stack without a head yet)
      float stack_kids_of_b = 0;
      for (int32_t k = kids_off[o + b]; k < kids_off[o + b + 1]; k++) {
        int32_t c = kids[k];
        if (on_stack[o + c] && head[o + c] == -1) stack_kids_of_b += 1;
      }
      c_shift += stack_kids_of_b;
      if (v0 >= 0) {
        c_reduce = (float)gold_kids_buf[o + v0];
        c_la = c_reduce;
        if (gh[v0] >= 0 && gh[v0] > b) c_la += 1;
        if (gh[b] >= 0 && gh[b] != v0 && (ghb_on_stack || gh[b] > b)) c_ra += 1;
        c_ra += stack_kids_of_b;
      }
    } else if (v0 >= 0) {
      c_reduce = (float)gold_kids_buf[o + v0];
    }
    // BREAK oracle terms: an unmarked gold boundary at B0 penalizes the
    // moves that pull B0 into the current sentence (SHIFT / RIGHT-ARC);
    // BREAK itself additionally pays for every gold arc it would sever
    // (stack tokens whose gold head or gold children lie at/after B0).
    bool at_gold_break = use_break && !gold_sent.empty() && b > 0 &&
                         gold_sent[o + b] == 1 && !sent_out[o + b];
    if (at_gold_break) {
      c_shift += 1;
      c_ra += 1;
    }
    r[0] = v[0] ? c_shift : KInvalid;
    r[1] = v[1] ? c_reduce : KInvalid;
    for (int32_t l = 0; l < n_labels; l++) {
      float la = c_la, ra = c_ra;
      if (b >= 0 && v0 >= 0) {
        if (gh[v0] == b && gl[v0] != l) la += 1;
        if (gh[b] == v0 && gl[b] != l) ra += 1;
      }
      r[2 + l] = v[2 + l] ? la : KInvalid;
      r[2 + n_labels + l] = v[2 + n_labels + l] ? ra : KInvalid;
    }
    if (use_break) {
      const int32_t bk = 2 + 2 * n_labels;
      float c_break = at_gold_break ? 0.f : 1.f;
      if (v[bk] && !gold_sent.empty()) {
        const int32_t* stk = stack.data() + o;
        for (int32_t k = 0; k < ssize[d]; k++) {
          int32_t s = stk[k];
          if (head[o + s] == -1 && gh[s] >= b) c_break += 1;  // head severed
          c_break += (float)gold_kids_buf[o + s];  // children severed
        }
      }
      r[bk] = v[bk] ? c_break : KInvalid;
    }
  }

  // Goldberg&Nivre dynamic-oracle costs; invalid actions get KInvalid.
  py::array_t<float> costs() const {
    if (!has_gold) throw std::runtime_error("costs() requires set_gold()");
    py::ssize_t A = n_actions();
    py::array_t<float> out({(py::ssize_t)n_docs, A});
    auto r = out.mutable_unchecked<2>();
    std::vector<uint8_t> v((size_t)A);
    for (int64_t i = 0; i < n_docs; i++) fill_costs(i, v.data(), r.mutable_data(i, 0));
    return out;
  }

  // ---- StepBatchIface (consumed by the _srx_hip C++ step-loop driver)
  int64_t n_states() const override { return n_docs; }
  int n_feats() const override { return 13; }
  int n_acts() const override { return (int)n_actions(); }
  int64_t max_transitions() const override {
    // buf advances exactly len times (SHIFT/RIGHT-ARC); each push is popped
    // at most once (REDUCE/LEFT-ARC); BREAK marks each token at most once
    // => <= (2|3)*len transitions per doc.
    return (use_break ? 3 : 2) * total;
  }

  int64_t pack_step(bool with_gold, int64_t pad_row, int32_t* act_idx,
                    int64_t* feats, uint8_t* valid_a, uint8_t* gold_a) override {
    int64_t Sa = 0;
    for (int64_t i = 0; i < n_docs; i++)
      if (!final_state(i)) act_idx[Sa++] = (int32_t)i;
    const int64_t A = n_actions();
    const int32_t L = n_labels;
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (int64_t k = 0; k < Sa; k++) {
      int32_t f32[13];
      int64_t i = act_idx[k];
      fill_features(i, f32);
      int64_t* fo = feats + k * 13;
      for (int q = 0; q < 13; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a + k * A;
      fill_valid(v, i);
      if (!with_gold) continue;
      // Min-cost gold mask WITHOUT materializing the per-action cost row:
      // all labeled LEFT-ARCs share cost c_la except the gold label (+1
      // for a wrong label only when the arc itself matches gold), same
      // for RIGHT-ARC — so 5 scalars + the gold labels determine the mask
      // and the label ranges become memsets (the 82-float crow + two
      // 82-wide scans were ~half the pack cost at 1M words).
      uint8_t* g = gold_a + k * A;
      scalar_costs(i, v, g, A, L);
    }
    return Sa;
  }

  // Shared scalar-cost core for pack_step: computes c_shift/c_reduce/
  // c_la/c_ra(/c_break) exactly as fill_costs does, then writes the
  // min-cost mask g[0..A) directly.
  inline void scalar_costs(int64_t d, const uint8_t* v, uint8_t* g,
                           int64_t A, int32_t L) const {
    const int64_t o = off[d];
    const int32_t* gh = gold_head.data() + o;
    const int32_t* gl = gold_label.data() + o;
    int32_t b = buf[d] < len[d] ? buf[d] : -1;
    int32_t v0 = s0(d);
    float c_shift = 0, c_reduce = 0, c_la = 0, c_ra = 0;
    if (b >= 0) {
      bool ghb_on_stack = gh[b] >= 0 && gh[b] < len[d] && on_stack[o + gh[b]];
      if (ghb_on_stack) c_shift += 1;
      float stack_kids_of_b = 0;
      for (int32_t k = kids_off[o + b]; k < kids_off[o + b + 1]; k++) {
        int32_t c = kids[k];
        if (on_stack[o + c] && head[o + c] == -1) stack_kids_of_b += 1;
      }
      c_shift += stack_kids_of_b;
      if (v0 >= 0) {
        c_reduce = (float)gold_kids_buf[o + v0];
        c_la = c_reduce;
        if (gh[v0] >= 0 && gh[v0] > b) c_la += 1;
        if (gh[b] >= 0 && gh[b] != v0 && (ghb_on_stack || gh[b] > b)) c_ra += 1;
        c_ra += stack_kids_of_b;
      }
    } else if (v0 >= 0) {
      c_reduce = (float)gold_kids_buf[o + v0];
    }
    bool at_gold_break = use_break && !gold_sent.empty() && b > 0 &&
                         gold_sent[o + b] == 1 && !sent_out[o + b];
    if (at_gold_break) {
      c_shift += 1;
      c_ra += 1;
    }
    // gold label deltas: LA(l) costs c_la + (arc matches gold but l wrong);
    // the +1 applies only when gh[v0] == b (resp. gh[b] == v0)
    int32_t la_gold = (b >= 0 && v0 >= 0 && gh[v0] == b) ? gl[v0] : -1;
    int32_t ra_gold = (b >= 0 && v0 >= 0 && gh[b] == v0) ? gl[b] : -1;
    const int32_t bk = 2 + 2 * L;
    float c_break = KInvalid;
    if (use_break && v[bk]) {
      c_break = at_gold_break ? 0.f : 1.f;
      if (!gold_sent.empty()) {
        const int32_t* stk = stack.data() + o;
        for (int32_t k2 = 0; k2 < ssize[d]; k2++) {
          int32_t s = stk[k2];
          if (head[o + s] == -1 && gh[s] >= (b >= 0 ? b : len[d])) c_break += 1;
          c_break += (float)gold_kids_buf[o + s];
        }
      }
    }
    float cmin = KInvalid;
    if (v[0]) cmin = std::min(cmin, c_shift);
    if (v[1]) cmin = std::min(cmin, c_reduce);
    if (v[2]) cmin = std::min(cmin, c_la);          // min over LA labels = c_la
    if (v[2 + L]) cmin = std::min(cmin, c_ra);      // min over RA labels = c_ra
    if (use_break && v[bk]) cmin = std::min(cmin, c_break);
    const float eps = 1e-6f;
    std::memset(g, 0, (size_t)A);
    if (v[0] && c_shift <= cmin + eps) g[0] = 1;
    if (v[1] && c_reduce <= cmin + eps) g[1] = 1;
    if (v[2]) {  // LA labels valid as a block (fill_valid sets all-or-none)
      if (c_la <= cmin + eps) {
        if (la_gold >= 0) {
          if (c_la + 1 <= cmin + eps) std::memset(g + 2, 1, (size_t)L);
          g[2 + la_gold] = 1;
        } else {
          std::memset(g + 2, 1, (size_t)L);
        }
      }
    }
    if (v[2 + L]) {
      if (c_ra <= cmin + eps) {
        if (ra_gold >= 0) {
          if (c_ra + 1 <= cmin + eps) std::memset(g + 2 + L, 1, (size_t)L);
          g[2 + L + ra_gold] = 1;
        } else {
          std::memset(g + 2 + L, 1, (size_t)L);
        }
      }
    }
    if (use_break && v[bk] && c_break <= cmin + eps) g[bk] = 1;
  }

  inline void leave_buffer(int64_t d, int32_t tok) {
    // token `tok` moves out of the buffer: its gold head loses one
    // in-buffer child (the c_reduce counter)
    if (has_gold) {
      int32_t h = gold_head[off[d] + tok];
      if (h >= 0 && h < len[d]) gold_kids_buf[off[d] + h]--;
    }
  }

  inline void apply_action(int64_t d, int32_t act) {
    const int64_t o = off[d];
    if (use_break && act == 2 + 2 * n_labels) {  // BREAK: mark B0 sent start
      sent_out[o + buf[d]] = 1;
      if (ssize[d] > 0) pending_break[d] = 1;  // cleanup pops follow
      return;
    }
    if (use_break && act == 1 && pending_break[d] && ssize[d] <= 1)
      pending_break[d] = 0;  // this pop empties the stack
    if (act == 0) {  // SHIFT
      on_stack[o + buf[d]] = 1;
      leave_buffer(d, buf[d]);
      stack[o + ssize[d]++] = buf[d];
      buf[d] += 1;
    } else if (act == 1) {  // REDUCE
      on_stack[o + stack[o + ssize[d] - 1]] = 0;
      ssize[d] -= 1;
    } else if (act < 2 + n_labels) {  // LEFT-ARC
      int32_t l = act - 2;
      int32_t v0 = stack[o + ssize[d] - 1];
      add_arc(d, buf[d], v0, l);
      on_stack[o + v0] = 0;
      ssize[d] -= 1;
    } else {  // RIGHT-ARC
      int32_t l = act - 2 - n_labels;
      int32_t v0 = stack[o + ssize[d] - 1];
      add_arc(d, v0, buf[d], l);
      on_stack[o + buf[d]] = 1;
      leave_buffer(d, buf[d]);
      stack[o + ssize[d]++] = buf[d];
      buf[d] += 1;
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
      apply_action(act_idx[k], actions[k]);
    }
  }

  // fused per-step call: ONE crossing of the pybind boundary returns
  // (active_idx, features, valid, is_gold) for the active states only.
  py::tuple step_arrays(bool with_gold, int64_t pad_row = -1) {
    const py::ssize_t A = n_actions();
    py::array_t<int32_t> act_full((py::ssize_t)n_docs);
    std::vector<int64_t> feats_tmp((size_t)n_docs * 13);
    std::vector<uint8_t> valid_tmp((size_t)n_docs * A);
    std::vector<uint8_t> gold_tmp(with_gold ? (size_t)n_docs * A : 0);
    int64_t Sa = pack_step(with_gold, pad_row, act_full.mutable_data(),
                           feats_tmp.data(), valid_tmp.data(), gold_tmp.data());
    py::array_t<int32_t> act((py::ssize_t)Sa);
    py::array_t<int64_t> feats({(py::ssize_t)Sa, (py::ssize_t)13});
    py::array_t<uint8_t> valid_a({(py::ssize_t)Sa, A});
    py::array_t<uint8_t> gold_a({with_gold ? (py::ssize_t)Sa : 0, A});
    std::memcpy(act.mutable_data(), act_full.data(), Sa * 4);
    std::memcpy(feats.mutable_data(), feats_tmp.data(), (size_t)Sa * 13 * 8);
    std::memcpy(valid_a.mutable_data(), valid_tmp.data(), (size_t)Sa * A);
    if (with_gold)
      std::memcpy(gold_a.mutable_data(), gold_tmp.data(), (size_t)Sa * A);
    return py::make_tuple(act, feats, valid_a, gold_a);
  }

  // Packed variant: ONE buffer = [feats int64 Sa*13][valid u8 Sa*A]
  // [gold u8 Sa*A (train only)] so the python side does a single H2D copy
  // per transition step.
  py::tuple step_arrays_packed(bool with_gold, int64_t pad_row = -1) {
    int64_t Sa0 = 0;
    for (int64_t i = 0; i < n_docs; i++)
      if (!final_state(i)) Sa0++;
    const py::ssize_t A = n_actions();
    const size_t fbytes = (size_t)Sa0 * 13 * 8;
    const size_t vbytes = (size_t)Sa0 * A;
    py::array_t<int32_t> act((py::ssize_t)Sa0);
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
      if (final_state(i)) continue;
      int32_t act = a(i);
      if (act < 0) continue;  // explicit no-op (already-final slot)
      apply_action(i, act);
    }
  }

  int64_t handle() { return (int64_t)(intptr_t)static_cast<srx::StepBatchIface*>(this); }

  py::array_t<int32_t> heads() const {
    py::array_t<int32_t> out((py::ssize_t)total);
    std::memcpy(out.mutable_data(), head.data(), total * sizeof(int32_t));
    return out;
  }

  py::array_t<int32_t> labels() const {
    py::array_t<int32_t> out((py::ssize_t)total);
    std::memcpy(out.mutable_data(), label.data(), total * sizeof(int32_t));
    return out;
  }

  py::array_t<int32_t> sent_starts() const {
    py::array_t<int32_t> out((py::ssize_t)total);
    std::memcpy(out.mutable_data(), sent_out.data(), total * sizeof(int32_t));
    return out;
  }
};

// --------------------------------------------------------------------- NER
// Per-token gold codes: 0 = O; for type t: 1+4t=B, 2+4t=I, 3+4t=L, 4+4t=U;
// -1 = MISSING ('-' in spaCy): no supervision for the token.
// Actions: 0 = OUT; for type t: 1+4t=BEGIN, 2+4t=IN, 3+4t=LAST, 4+4t=UNIT.
struct BiluoBatch : public srx::StepBatchIface {
  int32_t n_types;
  int64_t n_docs = 0, total = 0;
  int32_t base_offset = 0;
  std::vector<int32_t> off;        // [n_docs + 1]
  std::vector<int32_t> len;        // [n_docs]
  std::vector<int32_t> cur;        // [n_docs] current token
  std::vector<int32_t> open;       // [n_docs] open entity type or -1
  std::vector<int32_t> open_start; // [n_docs]
  std::vector<int32_t> tags;       // flat [total] emitted action codes
  std::vector<int32_t> gold;       // flat [total]
  bool has_gold = false;

  BiluoBatch(py::array_t<int32_t, py::array::c_style | py::array::forcecast> lengths,
             int32_t n_types_, int32_t base_offset_ = 0)
      : n_types(n_types_), base_offset(base_offset_) {
    auto L = lengths.unchecked<1>();
    n_docs = L.shape(0);
    off.resize(n_docs + 1);
    len.resize(n_docs);
    off[0] = 0;
    for (int64_t i = 0; i < n_docs; i++) {
      len[i] = L(i);
      off[i + 1] = off[i] + L(i);
    }
    total = off[n_docs];
    cur.assign(n_docs, 0);
    open.assign(n_docs, -1);
    open_start.assign(n_docs, -1);
    tags.assign(total, 0);
  }

  void set_gold(py::array_t<int32_t, py::array::c_style | py::array::forcecast> codes) {
    // global gold codes; this shard's slice starts at base_offset
    auto G = codes.unchecked<1>();
    if ((int64_t)G.shape(0) < base_offset + total)
      throw std::runtime_error("set_gold: codes shorter than batch");
    gold.resize(total);
    std::memcpy(gold.data(), G.data(0) + base_offset, total * sizeof(int32_t));
    has_gold = true;
  }

  int32_t n_actions() const { return 1 + 4 * n_types; }
  size_t size() const { return (size_t)n_docs; }

  inline bool final_state(int64_t d) const { return cur[d] >= len[d]; }

  py::array_t<uint8_t> is_final() const {
    py::array_t<uint8_t> out((py::ssize_t)n_docs);
    auto r = out.mutable_unchecked<1>();
    for (int64_t i = 0; i < n_docs; i++) r(i) = final_state(i) ? 1 : 0;
    return out;
  }

  // 6 context tokens: [i-2, i-1, i, i+1, i+2, open_start]  (batch-flat, -1 pad)
  void fill_features(int64_t d, int32_t* out) const {
    int32_t i = cur[d];
    int32_t f[6] = {i - 2, i - 1, i, i + 1, i + 2, open_start[d]};
    const int32_t ob = base_offset + off[d];
    for (int k = 0; k < 6; k++)
      out[k] = (f[k] >= 0 && f[k] < len[d]) ? ob + f[k] : -1;
  }

  py::array_t<int32_t> features() const {
    py::array_t<int32_t> out({(py::ssize_t)n_docs, (py::ssize_t)6});
    auto r = out.mutable_unchecked<2>();
    for (int64_t s = 0; s < n_docs; s++) fill_features(s, r.mutable_data(s, 0));
    return out;
  }

  void fill_valid(uint8_t* v, int64_t d) const {
    const int32_t A = n_actions();
    std::fill(v, v + A, 0);
    if (final_state(d)) return;
    bool last_tok = cur[d] == len[d] - 1;
    if (open[d] < 0) {
      v[0] = 1;  // OUT
      for (int32_t t = 0; t < n_types; t++) {
        if (!last_tok) v[1 + 4 * t] = 1;  // BEGIN needs a following token
        v[4 + 4 * t] = 1;                 // UNIT
      }
    } else {
      if (!last_tok) v[2 + 4 * open[d]] = 1;  // IN
      v[3 + 4 * open[d]] = 1;                 // LAST
    }
  }

  py::array_t<uint8_t> valid() const {
    py::ssize_t A = n_actions();
    py::array_t<uint8_t> out({(py::ssize_t)n_docs, A});
    auto r = out.mutable_unchecked<2>();
    for (int64_t i = 0; i < n_docs; i++) fill_valid(r.mutable_data(i, 0), i);
    return out;
  }

  py::array_t<float> costs() const {
    if (!has_gold) throw std::runtime_error("costs() requires set_gold()");
    py::ssize_t A = n_actions();
    py::array_t<float> out({(py::ssize_t)n_docs, A});
    auto r = out.mutable_unchecked<2>();
    std::vector<uint8_t> v((size_t)A);
    for (int64_t s = 0; s < n_docs; s++) {
      fill_valid(v.data(), s);
      for (py::ssize_t a = 0; a < A; a++) {
        if (!v[a]) { r(s, a) = KInvalid; continue; }
        int32_t g = final_state(s) ? -2 : gold[off[s] + cur[s]];
        // g == -1 (missing annotation): every valid action is free
        r(s, a) = (g == -1 || (int32_t)a == g) ? 0.0f : 1.0f;
      }
    }
    return out;
  }

  // ---- StepBatchIface
  int64_t n_states() const override { return n_docs; }
  int n_feats() const override { return 6; }
  int n_acts() const override { return (int)n_actions(); }
  int64_t max_transitions() const override { return total; }  // 1 action/token

  int64_t pack_step(bool with_gold, int64_t pad_row, int32_t* act_idx,
                    int64_t* feats, uint8_t* valid_a, uint8_t* gold_a) override {
    int64_t Sa = 0;
    for (int64_t i = 0; i < n_docs; i++)
      if (!final_state(i)) act_idx[Sa++] = (int32_t)i;
    const int64_t A = n_actions();
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (Sa > 512)
#endif
    for (int64_t k = 0; k < Sa; k++) {
      int64_t i = act_idx[k];
      int32_t f32[6];
      fill_features(i, f32);
      int64_t* fo = feats + k * 6;
      for (int q = 0; q < 6; q++) fo[q] = f32[q] < 0 ? pad_row : (int64_t)f32[q];
      uint8_t* v = valid_a + k * A;
      fill_valid(v, i);
      if (with_gold) {
        uint8_t* g = gold_a + k * A;
        int32_t gcode = final_state(i) ? -2 : gold[off[i] + cur[i]];
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

  inline void apply_action(int64_t d, int32_t act) {
    tags[off[d] + cur[d]] = act;
    if (act == 0) {
      open[d] = -1; open_start[d] = -1;
    } else {
      int32_t t = (act - 1) / 4;
      int32_t kind = (act - 1) % 4;  // 0=B,1=I,2=L,3=U
      if (kind == 0) { open[d] = t; open_start[d] = cur[d]; }
      else if (kind == 1) { /* stays open */ }
      else { open[d] = -1; open_start[d] = -1; }
    }
    cur[d] += 1;
  }

  void advance_active(const int32_t* act_idx, const int32_t* actions,
                      int64_t n) override {
#ifdef _OPENMP
#pragma omp parallel for num_threads(srx_nthreads()) schedule(static) if (n > 2048)
#endif
    for (int64_t k = 0; k < n; k++) {
      if (actions[k] < 0) continue;
      apply_action(act_idx[k], actions[k]);
    }
  }

  py::tuple step_arrays(bool with_gold, int64_t pad_row = -1) {
    const py::ssize_t A = n_actions();
    py::array_t<int32_t> act_full((py::ssize_t)n_docs);
    std::vector<int64_t> feats_tmp((size_t)n_docs * 6);
    std::vector<uint8_t> valid_tmp((size_t)n_docs * A);
    std::vector<uint8_t> gold_tmp(with_gold ? (size_t)n_docs * A : 0);
    int64_t Sa = pack_step(with_gold, pad_row, act_full.mutable_data(),
                           feats_tmp.data(), valid_tmp.data(), gold_tmp.data());
    py::array_t<int32_t> act((py::ssize_t)Sa);
    py::array_t<int64_t> feats({(py::ssize_t)Sa, (py::ssize_t)6});
    py::array_t<uint8_t> valid_a({(py::ssize_t)Sa, A});
    py::array_t<uint8_t> gold_a({with_gold ? (py::ssize_t)Sa : 0, A});
    std::memcpy(act.mutable_data(), act_full.data(), Sa * 4);
    std::memcpy(feats.mutable_data(), feats_tmp.data(), (size_t)Sa * 6 * 8);
    std::memcpy(valid_a.mutable_data(), valid_tmp.data(), (size_t)Sa * A);
    if (with_gold)
      std::memcpy(gold_a.mutable_data(), gold_tmp.data(), (size_t)Sa * A);
    return py::make_tuple(act, feats, valid_a, gold_a);
  }

  py::tuple step_arrays_packed(bool with_gold, int64_t pad_row = -1) {
    int64_t Sa0 = 0;
    for (int64_t i = 0; i < n_docs; i++)
      if (!final_state(i)) Sa0++;
    const py::ssize_t A = n_actions();
    const size_t fbytes = (size_t)Sa0 * 6 * 8;
    const size_t vbytes = (size_t)Sa0 * A;
    py::array_t<int32_t> act((py::ssize_t)Sa0);
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
      if (final_state(s)) continue;
      int32_t act = a(s);
      if (act < 0) continue;
      apply_action(s, act);
    }
  }

  int64_t handle() { return (int64_t)(intptr_t)static_cast<srx::StepBatchIface*>(this); }

  py::array_t<int32_t> tags_out() const {
    py::array_t<int32_t> out((py::ssize_t)total);
    std::memcpy(out.mutable_data(), tags.data(), total * sizeof(int32_t));
    return out;
  }
};

}  // namespace

void init_transitions(py::module_& m) {
  py::class_<ArcEagerBatch>(m, "ArcEagerBatch")
      .def(py::init<py::array_t<int32_t, py::array::c_style | py::array::forcecast>, int32_t, int32_t, bool>(),
           py::arg("lengths"), py::arg("n_labels"), py::arg("base_offset") = 0,
           py::arg("use_break") = false)
      .def("set_gold", &ArcEagerBatch::set_gold, py::arg("heads"), py::arg("labels"))
      .def("set_sent_gold", &ArcEagerBatch::set_sent_gold, py::arg("sent_starts"))
      .def("sent_starts", &ArcEagerBatch::sent_starts)
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
      .def("tags", &BiluoBatch::tags_out);
}
