// Virtual seam between the CPU transition systems (_srx_cpu,
// transitions.cpp, compiled with g++) and the GPU step-loop driver
// (_srx_hip, srx_steploop.hip, compiled with hipcc).  Both sides are
// Itanium-ABI C++ on libstdc++, so a vtable pointer passed as a uintptr_t
// through Python resolves correctly across the two shared objects.
//
// The contract exists so the WHOLE per-batch transition loop (pack step ->
// H2D -> score kernels -> D2H actions -> advance) can run in native code
// with one Python crossing per batch instead of ~10 per transition step
// (round-1 profile: ~275 ms/step wall vs 79 ms GPU at 512k words — the
// host-side loop was the bottleneck).
#pragma once
#include <cstdint>

namespace srx {

struct StepBatchIface {
  virtual ~StepBatchIface() = default;

  // Number of states (docs) in the batch.
  virtual int64_t n_states() const = 0;
  // Feature slots per state (13 parser / 6 NER).
  virtual int n_feats() const = 0;
  // Action-space size.
  virtual int n_acts() const = 0;
  // Upper bound on the total number of (state, step) scoring rows over the
  // whole loop (arena capacity): 2*total_tokens for arc-eager, total_tokens
  // for BILUO.
  virtual int64_t max_transitions() const = 0;

  // Pack the ACTIVE states' step data into caller-owned buffers:
  //   act_idx [n_states()]      int32  — indices of active states
  //   feats   [n_states()*nF]   int64  — missing slots remapped to pad_row
  //   valid   [n_states()*A]    uint8
  //   gold    [n_states()*A]    uint8  — only written when with_gold
  // Returns Sa = number of active states (0 => loop finished).
  virtual int64_t pack_step(bool with_gold, int64_t pad_row, int32_t* act_idx,
                            int64_t* feats, uint8_t* valid, uint8_t* gold) = 0;

  // Apply actions[k] to states[act_idx[k]] for k in [0, n).
  virtual void advance_active(const int32_t* act_idx, const int32_t* actions,
                              int64_t n) = 0;
};

}  // namespace srx
