// C++-owned transition step loop.
//
// Round 1 ran the parser/NER greedy loop from Python: ~10 python/pybind
// crossings + a torch-dispatch addmm + 2 kernel launches + a blocking D2H
// per transition step, ~121 steps x 2 pipes per training batch — the wall
// clock was ~275 ms/step against 79 ms of GPU work.  This TU owns the WHOLE
// loop natively: per step it packs the active states (OpenMP, directly into
// pinned staging), uploads one async copy per array, launches ONE fused
// kernel (gather+sum+bias+maxout + upper GEMM from LDS + masked argmax),
// copies [Sa] int32 actions back, and advances the C++ state machine —
// interleaving multiple units (parser/NER x shards) so one unit's CPU
// phase hides under another's GPU phase.  One Python crossing per BATCH.
//
// Training outputs land in device arenas (scores/gold/valid/feats/which/
// hidden over ALL steps) so the loss + backward run BATCHED afterwards:
// one fused CE kernel + 3 large GEMMs + one scatter for the whole loop
// instead of per-step autograd nodes (replaces fusedstep:: in srx_ext.hip).
//
// Contract mirrored from spaCy's parser_model.pyx step semantics
// (SURVEY.md §2.2 N8); state machines via srx::StepBatchIface
// (ops/csrc/step_iface.h, implemented by transitions.cpp across the
// .so boundary).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDACachingAllocator.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <mutex>
#include <stdexcept>
#include <thread>
#include <tuple>
#include <unordered_map>
#include <vector>

#include "../csrc/step_iface.h"
#include "srx_common.hip.h"

namespace {

// ------------------------------------------------------------ pinned pool
// hipHostMalloc is ~ms-scale; staging buffers are cached forever and reused
// across steps (bench/training reuse identical geometries).
struct PinnedPool {
  std::mutex mu;
  std::unordered_map<size_t, std::vector<void*>> free_;

  static size_t round_sz(size_t b) {
    size_t q = 1 << 20;
    return ((b + q - 1) / q) * q;
  }

  void* get(size_t bytes) {
    size_t sz = round_sz(bytes);
    {
      std::lock_guard<std::mutex> lock(mu);
      auto it = free_.find(sz);
      if (it != free_.end() && !it->second.empty()) {
        void* p = it->second.back();
        it->second.pop_back();
        return p;
      }
    }
    void* p = nullptr;
    hipError_t err = hipHostMalloc(&p, sz, hipHostMallocDefault);
    TORCH_CHECK(err == hipSuccess, "hipHostMalloc(", sz, ") failed: ",
                hipGetErrorString(err));
    return p;
  }

  void put(void* p, size_t bytes) {
    std::lock_guard<std::mutex> lock(mu);
    free_[round_sz(bytes)].push_back(p);
  }
};

PinnedPool& pool() {
  static PinnedPool p;
  return p;
}

// --------------------------------------------------------- fused step kernel
// One wave per state; upper weights staged TRANSPOSED in LDS ([h][A] layout:
// lane a reads Wlds[h*A+a] — consecutive lanes, conflict-free) once per
// block.  Phases per state:
//   1. gather nF precomputed rows, sum + bias, maxout(P=2) -> hidden
//      (written to arena + this wave's LDS slot)
//   2. scores[a] = upperB[a] + dot(hidden, upperW[a,:]) from LDS broadcast
//   3. masked argmax (sel_mask first, valid fallback) -> actions[s]
template <typename T>
__global__ void fused_step_all_kernel(
    const T* __restrict__ pre, const int64_t* __restrict__ feats,
    const T* __restrict__ lowerB, const T* __restrict__ upperW,
    const T* __restrict__ upperB, const uint8_t* __restrict__ sel_mask,
    const uint8_t* __restrict__ valid, T* __restrict__ hidden_out,
    uint8_t* __restrict__ which_out, T* __restrict__ scores_out,
    int32_t* __restrict__ actions_out, long S, int nF, int H, int A) {
  extern __shared__ char smem[];
  T* Wlds = (T*)smem;                       // [H][A] transposed
  float* Blds = (float*)(Wlds + (size_t)H * A);  // [A]
  float* hid_lds = Blds + A;                // [waves_per_block][H]
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const int wslot = threadIdx.x / SRX_WAVE;
  float* my_hid = hid_lds + (size_t)wslot * H;
  // block prologue: stage upper weights (transposed) + bias
  for (int idx = threadIdx.x; idx < A * H; idx += blockDim.x) {
    int a = idx / H, h = idx % H;
    Wlds[(size_t)h * A + a] = upperW[idx];
  }
  for (int a = threadIdx.x; a < A; a += blockDim.x)
    Blds[a] = Elem<T>::ld(upperB + a);
  __syncthreads();

  const int HP = 2 * H;
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long s = wave; s < S; s += nwaves) {
    const int64_t* fs = feats + s * nF;
    // phase 1: hidden
    for (int h = lane; h < H; h += SRX_WAVE) {
      float acc0 = Elem<T>::ld(lowerB + h);
      float acc1 = Elem<T>::ld(lowerB + H + h);
      for (int f = 0; f < nF; f++) {
        const T* row = pre + (fs[f] * (long)nF + f) * HP;
        acc0 += Elem<T>::ld(row + h);
        acc1 += Elem<T>::ld(row + H + h);
      }
      bool second = acc1 > acc0;
      float hv = second ? acc1 : acc0;
      Elem<T>::st(hidden_out + s * (long)H + h, hv);
      which_out[s * (long)H + h] = (uint8_t)second;
      my_hid[h] = hv;
    }
    // same wave reads its own LDS slot; lgkmcnt waits are compiler-inserted
    // phase 2 + 3: scores and masked argmax
    float bg = -1e38f, bv = -1e38f;
    int ig = INT32_MAX, iv = INT32_MAX;
    const uint8_t* grow = sel_mask + s * (long)A;
    const uint8_t* vrow = valid + s * (long)A;
    for (int a = lane; a < A; a += SRX_WAVE) {
      float acc = Blds[a];
      for (int h = 0; h < H; h++)
        acc += my_hid[h] * Elem<T>::ld(Wlds + (size_t)h * A + a);
      Elem<T>::st(scores_out + s * (long)A + a, acc);
      if (grow[a] && (acc > bg || (acc == bg && a < ig))) { bg = acc; ig = a; }
      if (vrow[a] && (acc > bv || (acc == bv && a < iv))) { bv = acc; iv = a; }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float obg = __shfl_xor(bg, off, SRX_WAVE);
      int oig = __shfl_xor(ig, off, SRX_WAVE);
      if (oig != INT32_MAX && (obg > bg || (obg == bg && oig < ig) || ig == INT32_MAX)) {
        bg = obg; ig = oig;
      }
      float obv = __shfl_xor(bv, off, SRX_WAVE);
      int oiv = __shfl_xor(iv, off, SRX_WAVE);
      if (oiv != INT32_MAX && (obv > bv || (obv == bv && oiv < iv) || iv == INT32_MAX)) {
        bv = obv; iv = oiv;
      }
    }
    if (lane == 0)
      actions_out[s] = ig != INT32_MAX ? ig : (iv != INT32_MAX ? iv : -1);
  }
}

constexpr int kBlockThreads = 256;  // 4 waves

template <typename T>
void launch_fused_step(const void* pre, const int64_t* feats, const void* lowerB,
                       const void* upperW, const void* upperB,
                       const uint8_t* sel, const uint8_t* valid, void* hidden,
                       uint8_t* which, void* scores, int32_t* actions, long S,
                       int nF, int H, int A, hipStream_t stream) {
  size_t lds = (size_t)H * A * sizeof(T) + A * sizeof(float) +
               (kBlockThreads / SRX_WAVE) * (size_t)H * sizeof(float);
  static std::once_flag attr_once;
  std::call_once(attr_once, [&]() {
    (void)hipFuncSetAttribute((const void*)fused_step_all_kernel<T>,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              128 * 1024);
  });
  long waves = S;
  int grid = (int)std::min<long>((waves * SRX_WAVE + kBlockThreads - 1) / kBlockThreads,
                                 16384);
  hipLaunchKernelGGL((fused_step_all_kernel<T>), dim3(grid), dim3(kBlockThreads),
                     lds, stream, (const T*)pre, feats, (const T*)lowerB,
                     (const T*)upperW, (const T*)upperB, sel, valid, (T*)hidden,
                     which, (T*)scores, actions, S, nF, H, A);
}

// ------------------------------------------------------------------- units
struct Unit {
  srx::StepBatchIface* b = nullptr;
  at::Tensor pre, lowerB, upperW, upperB;
  bool train = false;
  int nF = 0, A = 0, H = 0, HP = 0;
  long T = 0;  // pad row index (pre.size(0) - 1)
  long cap = 0, used = 0, nst = 0;
  at::Tensor feats_a, valid_a, gold_a, hidden_a, which_a, scores_a, actions_d;
  // pinned staging
  int32_t* act_idx_h = nullptr;
  int64_t* feats_h = nullptr;
  uint8_t* valid_h = nullptr;
  uint8_t* gold_h = nullptr;
  int32_t* actions_h = nullptr;
  size_t staging_bytes = 0;
  void* staging = nullptr;
  hipEvent_t ev = nullptr;
  bool done = false, pending = false;
  long Sa = 0;
  int steps = 0;
  // each unit runs on its OWN stream from torch's pool: on a shared stream
  // unit B's tiny D2H queues behind unit A's kernels, adding A's latency to
  // B's event — exactly the serialization the interleave is meant to hide.
  c10::cuda::CUDAStream stream = c10::cuda::getDefaultCUDAStream();
  hipStream_t hs = nullptr;
};

// Spin-wait: hipEventSynchronize parks the thread in the kernel driver and
// the wakeup costs ~50-200 us — per transition step, on the critical chain.
inline void spin_wait(hipEvent_t ev) {
  while (hipEventQuery(ev) == hipErrorNotReady) {
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
}

// SRX_LOOP_STATS=1: per-call phase accounting (stderr) — wait = event spin,
// pack = state-machine feature/cost extraction, gpu = copies+launch issue,
// adv = state advance.
struct LoopStats {
  double pack_ms = 0, wait_ms = 0, gpu_ms = 0, adv_ms = 0;
  long iters = 0;
};

inline double now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6;
}

using TaskArg = std::tuple<int64_t, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, bool>;

}  // namespace

// One python call per batch: run every unit's transition loop to completion,
// interleaved.  Returns per unit {scores, gold, valid, feats, which, hidden}
// arena slices over all steps (empty tensors for decode units).
std::vector<std::vector<at::Tensor>> srx_run_transition_loop(
    std::vector<TaskArg> tasks) {
  auto cur = at::cuda::getCurrentCUDAStream();
  // all unit streams wait for the current stream (pre/weights produced there)
  hipEvent_t start_ev;
  hipEventCreateWithFlags(&start_ev, hipEventDisableTiming);
  hipEventRecord(start_ev, cur.stream());
  std::vector<Unit> units(tasks.size());

  for (size_t i = 0; i < tasks.size(); i++) {
    Unit& u = units[i];
    u.b = (srx::StepBatchIface*)(intptr_t)std::get<0>(tasks[i]);
    u.pre = std::get<1>(tasks[i]);
    u.lowerB = std::get<2>(tasks[i]);
    u.upperW = std::get<3>(tasks[i]);
    u.upperB = std::get<4>(tasks[i]);
    u.train = std::get<5>(tasks[i]);
    TORCH_CHECK(u.pre.is_cuda() && u.pre.is_contiguous(), "pre must be contiguous CUDA");
    TORCH_CHECK(u.upperW.is_contiguous() && u.upperB.is_contiguous() &&
                u.lowerB.is_contiguous(), "weights must be contiguous");
    u.nF = u.b->n_feats();
    u.A = u.b->n_acts();
    u.HP = (int)u.pre.size(-1);
    u.H = u.HP / 2;
    u.T = u.pre.size(0) - 1;
    u.nst = u.b->n_states();
    TORCH_CHECK(u.upperW.size(0) == u.A && u.upperW.size(1) == u.H,
                "upperW shape mismatch");
    TORCH_CHECK((size_t)u.H * u.A * u.pre.element_size() <= 96 * 1024,
                "upper layer too large for LDS staging (H*A)");
    u.cap = u.train ? u.b->max_transitions() + 1 : u.nst;
    auto opt = u.pre.options();
    auto optb = opt.dtype(at::kByte);
    u.feats_a = at::empty({u.cap, (long)u.nF}, opt.dtype(at::kLong));
    u.valid_a = at::empty({u.cap, (long)u.A}, optb);
    u.gold_a = u.train ? at::empty({u.cap, (long)u.A}, optb) : u.valid_a;
    u.hidden_a = at::empty({u.cap, (long)u.H}, opt);
    u.which_a = at::empty({u.cap, (long)u.H}, optb);
    u.scores_a = at::empty({u.cap, (long)u.A}, opt);
    u.actions_d = at::empty({u.nst}, opt.dtype(at::kInt));
    // pinned staging layout: [act_idx i32][actions i32][feats i64][valid][gold]
    size_t bytes = (size_t)u.nst * (4 + 4 + (size_t)u.nF * 8 + 2 * (size_t)u.A) + 64;
    u.staging_bytes = bytes;
    u.staging = pool().get(bytes);
    char* p = (char*)u.staging;
    u.act_idx_h = (int32_t*)p;            p += (size_t)u.nst * 4;
    u.actions_h = (int32_t*)p;            p += (size_t)u.nst * 4;
    u.feats_h = (int64_t*)p;              p += (size_t)u.nst * u.nF * 8;
    u.valid_h = (uint8_t*)p;              p += (size_t)u.nst * u.A;
    u.gold_h = (uint8_t*)p;
    hipEventCreateWithFlags(&u.ev, hipEventDisableTiming);
    u.stream = c10::cuda::getStreamFromPool(false, u.pre.get_device());
    u.hs = u.stream.stream();
    hipStreamWaitEvent(u.hs, start_ev, 0);
    // caching-allocator safety: these tensors were allocated on the current
    // stream but are touched on the unit stream
    for (const at::Tensor* t : {&u.feats_a, &u.valid_a, &u.gold_a, &u.hidden_a,
                                &u.which_a, &u.scores_a, &u.actions_d, &u.pre,
                                &u.lowerB, &u.upperW, &u.upperB}) {
      c10::cuda::CUDACachingAllocator::recordStream(t->storage().data_ptr(),
                                                    u.stream);
    }
  }
  hipEventDestroy(start_ev);

  // one HOST THREAD per unit: each unit's serial chain (advance -> pack ->
  // H2D -> kernel -> D2H) runs independently on its own stream, so the
  // parser's CPU phases overlap the NER's GPU phases AND vice versa
  // (single-threaded round-robin serialized the pack work of all units:
  // pack was 27 ms/step of the 52 ms loop at 1M words).
  const long max_iters = 1L << 30;
  static const bool loop_stats = getenv("SRX_LOOP_STATS") != nullptr;
  LoopStats st;
  std::mutex st_mu;
  auto run_unit = [&](Unit& u) {
    LoopStats ls;
    double t0 = 0;
    long guard = 0;
    while (!u.done) {
      if (guard++ >= max_iters) throw std::runtime_error("loop stuck");
      if (u.pending) {
        if (loop_stats) t0 = now_ms();
        spin_wait(u.ev);
        if (loop_stats) { ls.wait_ms += now_ms() - t0; t0 = now_ms(); }
        u.b->advance_active(u.act_idx_h, u.actions_h, u.Sa);
        if (loop_stats) ls.adv_ms += now_ms() - t0;
        u.pending = false;
      }
      if (loop_stats) t0 = now_ms();
      long Sa = u.b->pack_step(u.train, u.T, u.act_idx_h, u.feats_h, u.valid_h,
                               u.gold_h);
      if (loop_stats) ls.pack_ms += now_ms() - t0;
      if (Sa == 0) {
        u.done = true;
        continue;
      }
      if (loop_stats) { ls.iters++; t0 = now_ms(); }
      long off = u.train ? u.used : 0;
      TORCH_CHECK(off + Sa <= u.cap, "transition arena overflow (", off, "+",
                  Sa, " > ", u.cap, ")");
      const size_t es = u.pre.element_size();
      char* feats_d = (char*)u.feats_a.data_ptr() + (size_t)off * u.nF * 8;
      char* valid_d = (char*)u.valid_a.data_ptr() + (size_t)off * u.A;
      char* gold_d = (char*)u.gold_a.data_ptr() + (size_t)off * u.A;
      char* hidden_d = (char*)u.hidden_a.data_ptr() + (size_t)off * u.H * es;
      uint8_t* which_d = (uint8_t*)u.which_a.data_ptr() + (size_t)off * u.H;
      char* scores_d = (char*)u.scores_a.data_ptr() + (size_t)off * u.A * es;
      int32_t* actions_d = (int32_t*)u.actions_d.data_ptr();
      hipMemcpyAsync(feats_d, u.feats_h, (size_t)Sa * u.nF * 8,
                     hipMemcpyHostToDevice, u.hs);
      hipMemcpyAsync(valid_d, u.valid_h, (size_t)Sa * u.A,
                     hipMemcpyHostToDevice, u.hs);
      if (u.train)
        hipMemcpyAsync(gold_d, u.gold_h, (size_t)Sa * u.A,
                       hipMemcpyHostToDevice, u.hs);
      const uint8_t* sel = u.train ? (const uint8_t*)gold_d : (const uint8_t*)valid_d;
      if (u.pre.scalar_type() == at::kBFloat16) {
        launch_fused_step<bf16_t>(u.pre.data_ptr(), (const int64_t*)feats_d,
                                  u.lowerB.data_ptr(), u.upperW.data_ptr(),
                                  u.upperB.data_ptr(), sel,
                                  (const uint8_t*)valid_d, hidden_d, which_d,
                                  scores_d, actions_d, Sa, u.nF, u.H, u.A, u.hs);
      } else {
        launch_fused_step<float>(u.pre.data_ptr(), (const int64_t*)feats_d,
                                 u.lowerB.data_ptr(), u.upperW.data_ptr(),
                                 u.upperB.data_ptr(), sel,
                                 (const uint8_t*)valid_d, hidden_d, which_d,
                                 scores_d, actions_d, Sa, u.nF, u.H, u.A, u.hs);
      }
      hipMemcpyAsync(u.actions_h, actions_d, (size_t)Sa * 4,
                     hipMemcpyDeviceToHost, u.hs);
      hipEventRecord(u.ev, u.hs);
      if (loop_stats) ls.gpu_ms += now_ms() - t0;
      u.pending = true;
      u.Sa = Sa;
      u.used += Sa;
      u.steps += 1;
    }
    if (loop_stats) {
      std::lock_guard<std::mutex> lock(st_mu);
      st.iters += ls.iters; st.pack_ms += ls.pack_ms; st.wait_ms += ls.wait_ms;
      st.gpu_ms += ls.gpu_ms; st.adv_ms += ls.adv_ms;
    }
  };
  if (units.size() <= 1) {
    for (Unit& u : units) run_unit(u);
  } else {
    std::vector<std::thread> threads;
    threads.reserve(units.size());
    std::vector<std::exception_ptr> errs(units.size());
    for (size_t i = 0; i < units.size(); i++)
      threads.emplace_back([&, i]() {
        try { run_unit(units[i]); }
        catch (...) { errs[i] = std::current_exception(); }
      });
    for (auto& t : threads) t.join();
    for (auto& e : errs)
      if (e) std::rethrow_exception(e);
  }
  if (loop_stats)
    fprintf(stderr,
            "[srx loop] iters=%ld pack=%.2fms wait=%.2fms issue=%.2fms adv=%.2fms\n",
            st.iters, st.pack_ms, st.wait_ms, st.gpu_ms, st.adv_ms);

  std::vector<std::vector<at::Tensor>> out;
  out.reserve(units.size());
  for (Unit& u : units) {
    hipEventDestroy(u.ev);
    pool().put(u.staging, u.staging_bytes);
    if (u.train) {
      long n = u.used;
      out.push_back({u.scores_a.narrow(0, 0, n), u.gold_a.narrow(0, 0, n),
                     u.valid_a.narrow(0, 0, n), u.feats_a.narrow(0, 0, n),
                     u.which_a.narrow(0, 0, n), u.hidden_a.narrow(0, 0, n)});
    } else {
      out.push_back({});
    }
  }
  return out;
}
