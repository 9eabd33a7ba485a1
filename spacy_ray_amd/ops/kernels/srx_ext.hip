// spacy_ray_amd._srx_hip — gfx950 kernel bindings (torch extension).
// Kernels live in the *.hip.h headers; this TU instantiates fp32 + bf16
// variants and exposes the op surface consumed by ops/api.py and
// parallel/engine.py.  Tested on-GPU against ops/torch_ref.py fp32.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <mutex>
#include <unordered_map>
#include <utility>
#include <vector>

#include "srx_common.hip.h"
#include "srx_elementwise.hip.h"
#include "srx_embed_parser.hip.h"
#include "srx_softmax_reduce.hip.h"
#include "srx_mwe.hip.h"
#include "srx_activations.hip.h"
#include "srx_attn.hip.h"

namespace {

constexpr int kBlock = 256;

inline int grid_for(long total, int per_thread = 1) {
  long blocks = (total + (long)kBlock * per_thread - 1) / ((long)kBlock * per_thread);
  return (int)std::min<long>(blocks, 16384);
}

inline void check_dev(const at::Tensor& t) {
  TORCH_CHECK(t.is_cuda(), "expected a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous");
}

#define DISPATCH_F(dtype, ...)                                        \
  if ((dtype) == at::kFloat) {                                        \
    using scalar_t = float;                                           \
    constexpr int kVec = 4;                                           \
    __VA_ARGS__;                                                      \
  } else if ((dtype) == at::kBFloat16) {                              \
    using scalar_t = bf16_t;                                          \
    constexpr int kVec = 8;                                           \
    __VA_ARGS__;                                                      \
  } else {                                                            \
    TORCH_CHECK(false, "unsupported dtype (need f32 or bf16)");       \
  }

// --------------------------------------------------------------- seq2col
// starts/ends: per-token doc-boundary uint8 masks, computed ONCE per batch
// on the python side (a masked_select/nonzero here would device-sync on
// every call — measured as a 33 ms/step stall in the encoder stack).
at::Tensor seq2col_fwd(at::Tensor X, at::Tensor starts, at::Tensor ends) {
  check_dev(X);
  long nT = X.size(0);
  int W = (int)X.size(1);
  auto Y = at::empty({nT, 3L * W}, X.options());
  if (nT == 0) return Y;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(X.scalar_type(), {
    const int V = (W % kVec == 0) ? kVec : 1;
    long total = nT * 3L * (W / V);
    if (V == kVec)
      hipLaunchKernelGGL((seq2col_fwd_kernel<scalar_t, kVec>), dim3(grid_for(total)),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)X.data_ptr(), (scalar_t*)Y.data_ptr(),
                         starts.data_ptr<uint8_t>(), ends.data_ptr<uint8_t>(), nT, W);
    else
      hipLaunchKernelGGL((seq2col_fwd_kernel<scalar_t, 1>), dim3(grid_for(nT * 3L * W)),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)X.data_ptr(), (scalar_t*)Y.data_ptr(),
                         starts.data_ptr<uint8_t>(), ends.data_ptr<uint8_t>(), nT, W);
  });
  return Y;
}

at::Tensor seq2col_bwd(at::Tensor dY, at::Tensor starts, at::Tensor ends,
                       c10::optional<at::Tensor> residual = c10::nullopt) {
  check_dev(dY);
  long nT = dY.size(0);
  int W3 = (int)dY.size(1);
  int W = W3 / 3;
  auto dX = at::empty({nT, (long)W}, dY.options());
  if (nT == 0) return dX;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(dY.scalar_type(), {
    const scalar_t* res =
        residual ? (const scalar_t*)residual->data_ptr() : nullptr;
    const int V = (W % kVec == 0) ? kVec : 1;
    if (V == kVec)
      hipLaunchKernelGGL((seq2col_bwd_kernel<scalar_t, kVec>),
                         dim3(grid_for(nT * (W / kVec))), dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), (scalar_t*)dX.data_ptr(),
                         starts.data_ptr<uint8_t>(), ends.data_ptr<uint8_t>(),
                         res, nT, W);
    else
      hipLaunchKernelGGL((seq2col_bwd_kernel<scalar_t, 1>),
                         dim3(grid_for(nT * (long)W)), dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), (scalar_t*)dX.data_ptr(),
                         starts.data_ptr<uint8_t>(), ends.data_ptr<uint8_t>(),
                         res, nT, W);
  });
  return dX;
}

// ---------------------------------------------------------------- maxout
std::vector<at::Tensor> maxout_fwd(at::Tensor X) {
  check_dev(X);
  int P = (int)X.size(-2);
  int W = (int)X.size(-1);
  long N = X.numel() / ((long)P * W);
  auto sizes = X.sizes().vec();
  sizes.erase(sizes.end() - 2);
  auto Y = at::empty(sizes, X.options());
  auto which = at::empty(sizes, X.options().dtype(at::kByte));
  if (N == 0) return {Y, which};
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(X.scalar_type(), {
    const int V = (W % kVec == 0) ? kVec : 1;
    if (V == kVec)
      hipLaunchKernelGGL((maxout_fwd_kernel<scalar_t, kVec>),
                         dim3(grid_for(N * (W / kVec))), dim3(kBlock), 0, stream,
                         (const scalar_t*)X.data_ptr(), (scalar_t*)Y.data_ptr(),
                         which.data_ptr<uint8_t>(), N, P, W);
    else
      hipLaunchKernelGGL((maxout_fwd_kernel<scalar_t, 1>),
                         dim3(grid_for(N * (long)W)), dim3(kBlock), 0, stream,
                         (const scalar_t*)X.data_ptr(), (scalar_t*)Y.data_ptr(),
                         which.data_ptr<uint8_t>(), N, P, W);
  });
  return {Y, which};
}

at::Tensor maxout_bwd(at::Tensor dY, at::Tensor which, long P) {
  check_dev(dY);
  int W = (int)dY.size(-1);
  long N = dY.numel() / W;
  auto sizes = dY.sizes().vec();
  sizes.insert(sizes.end() - 1, P);
  auto dX = at::empty(sizes, dY.options());
  if (N == 0) return dX;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(dY.scalar_type(), {
    const int V = (W % kVec == 0) ? kVec : 1;
    long total = N * P * (W / (V == kVec ? kVec : 1));
    if (V == kVec)
      hipLaunchKernelGGL((maxout_bwd_kernel<scalar_t, kVec>), dim3(grid_for(total)),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), which.data_ptr<uint8_t>(),
                         (scalar_t*)dX.data_ptr(), N, (int)P, W);
    else
      hipLaunchKernelGGL((maxout_bwd_kernel<scalar_t, 1>), dim3(grid_for(N * P * (long)W)),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), which.data_ptr<uint8_t>(),
                         (scalar_t*)dX.data_ptr(), N, (int)P, W);
  });
  return dX;
}

// ------------------------------------------------------------- layernorm
std::vector<at::Tensor> layernorm_fwd(at::Tensor X, at::Tensor g, at::Tensor b, double eps) {
  check_dev(X);
  int W = (int)X.size(-1);
  long N = X.numel() / W;
  auto Y = at::empty_like(X);
  auto mu = at::empty({N}, X.options().dtype(at::kFloat));
  auto rstd = at::empty({N}, X.options().dtype(at::kFloat));
  if (N == 0) return {Y, mu, rstd};
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(N * SRX_WAVE);
  DISPATCH_F(X.scalar_type(), {
    if (W % (2 * SRX_WAVE) == 0)
      hipLaunchKernelGGL((layernorm_fwd_v2_kernel<scalar_t>), dim3(grid), dim3(kBlock), 0,
                         stream, (const scalar_t*)X.data_ptr(),
                         (const scalar_t*)g.data_ptr(), (const scalar_t*)b.data_ptr(),
                         (scalar_t*)Y.data_ptr(), mu.data_ptr<float>(),
                         rstd.data_ptr<float>(), N, W, (float)eps);
    else
      hipLaunchKernelGGL((layernorm_fwd_kernel<scalar_t>), dim3(grid), dim3(kBlock), 0,
                         stream, (const scalar_t*)X.data_ptr(),
                         (const scalar_t*)g.data_ptr(), (const scalar_t*)b.data_ptr(),
                         (scalar_t*)Y.data_ptr(), mu.data_ptr<float>(),
                         rstd.data_ptr<float>(), N, W, (float)eps);
  });
  return {Y, mu, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dY, at::Tensor X, at::Tensor g,
                                      at::Tensor mu, at::Tensor rstd,
                                      bool deterministic = false) {
  check_dev(dY);
  int W = (int)X.size(-1);
  long N = X.numel() / W;
  auto dX = at::empty_like(X);
  auto acc_dt = deterministic ? at::kLong : at::kFloat;
  auto dg32 = at::zeros({W}, X.options().dtype(acc_dt));
  auto db32 = at::zeros({W}, X.options().dtype(acc_dt));
  auto stream = at::cuda::getCurrentCUDAStream();
  TORCH_CHECK(W <= SRX_LN_MAX_W, "layernorm width > ", SRX_LN_MAX_W);
  if (N > 0) {
    // cap the grid so each wave covers many rows: the dg/db column sums are
    // register-accumulated per wave with one atomic per column at the end.
    // Narrow layers (CNN W=96) keep a tight cap — extra blocks only
    // serialize on the few dg/db addresses; wide layers (trf W=768) spread
    // the atomics over 8x the columns and need more waves to fill 256 CUs
    // at large N (measured 2.0 ms/call at N=260k W=768 under the 1024 cap).
    // (re-measured r2: the old 1024 cap for narrow layers starved the
    // 1M-row encoder backward of parallelism — 0.73 ms/call at 10x off
    // bandwidth; per-wave register accumulation keeps the dg/db atomic
    // count at waves*W regardless, so more waves are safe)
    long cap = 4096;
    int grid = (int)std::min<long>((N + 3) / 4, cap);
    bool v2 = W % (2 * SRX_WAVE) == 0;
    DISPATCH_F(X.scalar_type(), {
      if (v2 && deterministic)
        hipLaunchKernelGGL((layernorm_bwd_v2_kernel<scalar_t, true>), dim3(grid),
                           dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                           (const scalar_t*)X.data_ptr(), (const scalar_t*)g.data_ptr(),
                           mu.data_ptr<float>(), rstd.data_ptr<float>(),
                           (scalar_t*)dX.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), N, W);
      else if (v2)
        hipLaunchKernelGGL((layernorm_bwd_v2_kernel<scalar_t, false>), dim3(grid),
                           dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                           (const scalar_t*)X.data_ptr(), (const scalar_t*)g.data_ptr(),
                           mu.data_ptr<float>(), rstd.data_ptr<float>(),
                           (scalar_t*)dX.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), N, W);
      else if (deterministic)
        hipLaunchKernelGGL((layernorm_bwd_kernel<scalar_t, true>), dim3(grid),
                           dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                           (const scalar_t*)X.data_ptr(), (const scalar_t*)g.data_ptr(),
                           mu.data_ptr<float>(), rstd.data_ptr<float>(),
                           (scalar_t*)dX.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), N, W);
      else
        hipLaunchKernelGGL((layernorm_bwd_kernel<scalar_t, false>), dim3(grid),
                           dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                           (const scalar_t*)X.data_ptr(), (const scalar_t*)g.data_ptr(),
                           mu.data_ptr<float>(), rstd.data_ptr<float>(),
                           (scalar_t*)dX.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), N, W);
    });
  }
  if (deterministic) {
    // fixed-point -> float (deterministic elementwise divide)
    auto s = 1.0 / 16777216.0;
    return {dX, (dg32.to(at::kFloat) * s).to(X.scalar_type()),
            (db32.to(at::kFloat) * s).to(X.scalar_type())};
  }
  return {dX, dg32.to(X.scalar_type()), db32.to(X.scalar_type())};
}

// ------------------------------------------------------------- hashembed
// out_/col_off: optional preallocated [nT, ldY] destination + column
// offset — the 4 attr tables write straight into their block of the
// concatenated embed matrix (no separate cat copy).
std::vector<at::Tensor> hashembed_fwd(at::Tensor table, at::Tensor ids, int64_t seed,
                                      c10::optional<at::Tensor> out_ = c10::nullopt,
                                      int64_t col_off = 0) {
  check_dev(table);
  check_dev(ids);
  TORCH_CHECK(ids.scalar_type() == at::kLong, "ids must be int64 (bit-cast uint64)");
  long nT = ids.size(0);
  int nrows = (int)table.size(0);
  int W = (int)table.size(1);
  at::Tensor Y;
  long ldY = W;
  char* yptr = nullptr;
  if (out_) {
    Y = *out_;
    TORCH_CHECK(Y.size(0) == nT && Y.scalar_type() == table.scalar_type());
    ldY = Y.size(1);
    yptr = (char*)Y.data_ptr() + col_off * Y.element_size();
  } else {
    Y = at::empty({nT, (long)W}, table.options());
    yptr = (char*)Y.data_ptr();
  }
  auto rows = at::empty({nT, 4}, table.options().dtype(at::kInt));
  if (nT == 0) return {Y, rows};
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(table.scalar_type(), {
    hipLaunchKernelGGL((hashembed_fwd_kernel<scalar_t>), dim3(grid_for(nT * SRX_WAVE)),
                       dim3(kBlock), 0, stream, (const scalar_t*)table.data_ptr(),
                       (const uint64_t*)ids.data_ptr<int64_t>(), (scalar_t*)yptr,
                       rows.data_ptr<int32_t>(), nT, nrows, W, ldY, (uint32_t)seed);
  });
  return {Y, rows};
}

at::Tensor hashembed_bwd(at::Tensor dY, at::Tensor rows, int64_t nrows) {
  check_dev(dY);
  long nT = dY.size(0);
  int W = (int)dY.size(1);
  auto dT32 = at::zeros({nrows, (long)W}, dY.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (nT > 0) {
    DISPATCH_F(dY.scalar_type(), {
      hipLaunchKernelGGL((hashembed_bwd_kernel<scalar_t>), dim3(grid_for(nT * SRX_WAVE)),
                         dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                         rows.data_ptr<int32_t>(), dT32.data_ptr<float>(), nT, W);
    });
  }
  return dT32.to(dY.scalar_type());
}

// ----------------------------------------------------- parser step score
std::vector<at::Tensor> parser_step_fwd(at::Tensor pre, at::Tensor feats, at::Tensor bias) {
  check_dev(pre);
  check_dev(feats);
  TORCH_CHECK(feats.scalar_type() == at::kLong, "feats must be int64");
  long S = feats.size(0);
  int nF = (int)feats.size(1);
  int HP = (int)pre.size(-1);
  int H = HP / 2;
  auto hidden = at::empty({S, (long)H}, pre.options());
  auto which = at::empty({S, (long)H}, pre.options().dtype(at::kByte));
  if (S == 0) return {hidden, which};
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(pre.scalar_type(), {
    hipLaunchKernelGGL((parser_step_fwd_kernel<scalar_t>), dim3(grid_for(S * SRX_WAVE)),
                       dim3(kBlock), 0, stream, (const scalar_t*)pre.data_ptr(),
                       feats.data_ptr<int64_t>(), (const scalar_t*)bias.data_ptr(),
                       (scalar_t*)hidden.data_ptr(), which.data_ptr<uint8_t>(), S, nF, H);
  });
  return {hidden, which};
}

// Accumulating variant: scatter this step's dPre gradient into ONE
// persistent fp32 buffer (no per-step [T+1,nF,HP] allocation/zeroing — the
// parser step loop calls this once per transition step; the caller zeroes
// the buffer once per batch and injects dPre back into autograd via a
// surrogate product).  Returns the per-step dBias (small, differentiable
// path for the bias parameter).
at::Tensor parser_step_bwd_into(at::Tensor dHidden, at::Tensor feats,
                                at::Tensor which, at::Tensor dPre32) {
  check_dev(dHidden);
  long S = feats.size(0);
  int nF = (int)feats.size(1);
  int HP = (int)dPre32.size(-1);
  int H = HP / 2;
  auto dBiasStep = at::zeros({(long)HP}, dHidden.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (S > 0) {
    DISPATCH_F(dHidden.scalar_type(), {
      hipLaunchKernelGGL((parser_step_bwd_kernel<scalar_t>),
                         dim3(grid_for(S * SRX_WAVE)), dim3(kBlock), 0, stream,
                         (const scalar_t*)dHidden.data_ptr(), feats.data_ptr<int64_t>(),
                         which.data_ptr<uint8_t>(), dPre32.data_ptr<float>(),
                         dBiasStep.data_ptr<float>(), S, nF, H);
    });
  }
  return dBiasStep.to(dHidden.scalar_type());
}

std::vector<at::Tensor> parser_step_bwd(at::Tensor dHidden, at::Tensor feats,
                                        at::Tensor which, int64_t T1, int64_t nF,
                                        int64_t HP) {
  check_dev(dHidden);
  long S = feats.size(0);
  int H = (int)HP / 2;
  auto dPre32 = at::zeros({T1, nF, HP}, dHidden.options().dtype(at::kFloat));
  auto dBias32 = at::zeros({HP}, dHidden.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (S > 0) {
    DISPATCH_F(dHidden.scalar_type(), {
      hipLaunchKernelGGL((parser_step_bwd_kernel<scalar_t>),
                         dim3(grid_for(S * SRX_WAVE)), dim3(kBlock), 0, stream,
                         (const scalar_t*)dHidden.data_ptr(), feats.data_ptr<int64_t>(),
                         which.data_ptr<uint8_t>(), dPre32.data_ptr<float>(),
                         dBias32.data_ptr<float>(), S, (int)nF, H);
    });
  }
  return {dPre32.to(dHidden.scalar_type()), dBias32.to(dHidden.scalar_type())};
}

// OUT (fp32 [No, W]) += segmented sums of SRC rows; dst_sorted must be sorted.
// OUT fp32: plain float atomics.  OUT int64: deterministic fixed-point
// (caller converts back with /2^24).
void seg_scatter_add(at::Tensor dst_sorted, at::Tensor src_idx, at::Tensor SRC,
                     at::Tensor OUT) {
  TORCH_CHECK(SRC.is_cuda() && SRC.dim() == 2 && SRC.stride(1) == 1,
              "SRC must be a CUDA row-view (innermost stride 1)");
  bool det = OUT.scalar_type() == at::kLong;
  TORCH_CHECK(det || OUT.scalar_type() == at::kFloat);
  TORCH_CHECK(dst_sorted.scalar_type() == at::kInt && src_idx.scalar_type() == at::kInt);
  long M = dst_sorted.numel();
  int W = (int)SRC.size(-1);
  long ldSRC = SRC.stride(0);
  TORCH_CHECK(W <= 1024, "seg_scatter_add W <= 1024");
  if (M == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  constexpr int CHUNK = 128;
  long waves = (M + CHUNK - 1) / CHUNK;
  int grid = (int)std::min<long>((waves * SRX_WAVE + kBlock - 1) / kBlock, 16384);
  DISPATCH_F(SRC.scalar_type(), {
    if (det)
      hipLaunchKernelGGL((seg_scatter_add_kernel<scalar_t, CHUNK, true>), dim3(grid),
                         dim3(kBlock), 0, stream, dst_sorted.data_ptr<int32_t>(),
                         src_idx.data_ptr<int32_t>(), (const scalar_t*)SRC.data_ptr(),
                         OUT.data_ptr(), M, W, ldSRC);
    else
      hipLaunchKernelGGL((seg_scatter_add_kernel<scalar_t, CHUNK, false>), dim3(grid),
                         dim3(kBlock), 0, stream, dst_sorted.data_ptr<int32_t>(),
                         src_idx.data_ptr<int32_t>(), (const scalar_t*)SRC.data_ptr(),
                         OUT.data_ptr(), M, W, ldSRC);
  });
}

at::Tensor action_select(at::Tensor scores, at::Tensor is_gold, at::Tensor valid) {
  check_dev(scores);
  long S = scores.size(0);
  int A = (int)scores.size(1);
  auto actions = at::empty({S}, scores.options().dtype(at::kInt));
  if (S == 0) return actions;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(scores.scalar_type(), {
    hipLaunchKernelGGL((action_select_kernel<scalar_t>), dim3(grid_for(S * SRX_WAVE)),
                       dim3(kBlock), 0, stream, (const scalar_t*)scores.data_ptr(),
                       is_gold.data_ptr<uint8_t>(), valid.data_ptr<uint8_t>(),
                       actions.data_ptr<int32_t>(), S, A);
  });
  return actions;
}

// ------------------------------------------------------------ fused Adam
// clip_scale: 0-dim fp32 DEVICE tensor (or empty for no clip) — keeping the
// scale on-device avoids a host sync per optimizer step.
void adam_step(at::Tensor grad, at::Tensor master, at::Tensor m, at::Tensor v,
               at::Tensor param_out, at::Tensor clip_scale, double lr, double beta1,
               double beta2, double eps, double wd, double bc1, double bc2) {
  check_dev(grad);
  long n = grad.numel();
  if (n == 0) return;
  const float* scale_ptr =
      clip_scale.numel() > 0 ? clip_scale.data_ptr<float>() : nullptr;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(grad.scalar_type(), {
    hipLaunchKernelGGL((adam_step_kernel<scalar_t>), dim3(grid_for(n, 4)), dim3(kBlock),
                       0, stream, (const scalar_t*)grad.data_ptr(),
                       master.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), (scalar_t*)param_out.data_ptr(), n,
                       scale_ptr, (float)lr, (float)beta1, (float)beta2,
                       (float)eps, (float)wd, (float)bc1, (float)bc2);
  });
}

// ------------------------------------------- transition-pipe batched ops
// Fused CE over ALL transition steps (the C++ loop's arenas): returns
// (loss_count fp32 [2], dScores) — dScores = softmax_over_valid - target,
// zero for invalid columns / unsupervised rows (see transition_ce_kernel).
std::vector<at::Tensor> transition_ce(at::Tensor scores, at::Tensor gold,
                                      at::Tensor valid,
                                      bool deterministic = false) {
  check_dev(scores);
  long N = scores.size(0);
  int A = (int)scores.size(1);
  auto dScores = at::empty_like(scores);
  auto loss = at::zeros({2}, scores.options().dtype(at::kFloat));
  auto colsum = at::zeros(
      {(long)A}, scores.options().dtype(deterministic ? at::kLong : at::kFloat));
  if (N == 0) return {loss, dScores, colsum.to(at::kFloat)};
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(scores.scalar_type(), {
    if (deterministic)
      hipLaunchKernelGGL((transition_ce_kernel<scalar_t, true>),
                         dim3(grid_for(N * SRX_WAVE)), dim3(kBlock), 0, stream,
                         (const scalar_t*)scores.data_ptr(),
                         gold.data_ptr<uint8_t>(), valid.data_ptr<uint8_t>(),
                         (scalar_t*)dScores.data_ptr(), loss.data_ptr<float>(),
                         colsum.data_ptr(), N, A);
    else
      hipLaunchKernelGGL((transition_ce_kernel<scalar_t, false>),
                         dim3(grid_for(N * SRX_WAVE)), dim3(kBlock), 0, stream,
                         (const scalar_t*)scores.data_ptr(),
                         gold.data_ptr<uint8_t>(), valid.data_ptr<uint8_t>(),
                         (scalar_t*)dScores.data_ptr(), loss.data_ptr<float>(),
                         colsum.data_ptr(), N, A);
  });
  if (deterministic)
    colsum = colsum.to(at::kFloat) * (1.0 / 16777216.0);
  return {loss, dScores, colsum};
}

// Batched dPre scatter over all steps: direct atomics for the near-uniform
// token destinations (packed-bf16 atomics when dPre is bf16 — halves the
// traffic and skips the fp32->bf16 convert of the whole buffer); the
// Zipf-hot PAD row and the bias column-sum are register-accumulated per
// wave into SEPARATE fp32 buffers.  Returns (dBias32 [HP], dPad32 [nF,HP]);
// the caller writes dPad into dPre's pad row.
std::vector<at::Tensor> dpre_scatter(at::Tensor dSummed, at::Tensor feats,
                                     at::Tensor dPre, int64_t pad_row) {
  check_dev(dSummed);
  TORCH_CHECK(feats.scalar_type() == at::kLong);
  long S = feats.size(0);
  int nF = (int)feats.size(1);
  int HP = (int)dPre.size(-1);
  TORCH_CHECK(HP <= 256, "dpre_scatter HP <= 256");
  TORCH_CHECK(nF <= 16, "dpre_scatter nF <= 16");
  int mode = dPre.scalar_type() == at::kBFloat16 ? 1
           : dPre.scalar_type() == at::kLong      ? 2
                                                  : 0;
  TORCH_CHECK(mode != 0 || dPre.scalar_type() == at::kFloat,
              "dPre must be bf16, fp32 or int64 (deterministic)");
  TORCH_CHECK(mode != 1 || HP % 2 == 0, "bf16 dPre needs even HP");
  auto acc_dt = mode == 2 ? at::kLong : at::kFloat;
  auto dBias32 = at::zeros({(long)HP}, dSummed.options().dtype(acc_dt));
  auto dPad32 = at::zeros({(long)nF, (long)HP}, dSummed.options().dtype(acc_dt));
  auto finish = [&](at::Tensor t) {
    return mode == 2 ? (t.to(at::kFloat) * (1.0 / 16777216.0)) : t;
  };
  if (S == 0) return {finish(dBias32), finish(dPad32)};
  auto stream = at::cuda::getCurrentCUDAStream();
  long waves = (S + 63) / 64;  // >=64 states per wave target
  int grid = (int)std::min<long>((waves * SRX_WAVE + kBlock - 1) / kBlock, 4096);
  grid = std::max(grid, 64);
  DISPATCH_F(dSummed.scalar_type(), {
    if (mode == 1)
      hipLaunchKernelGGL((dpre_scatter_kernel<scalar_t, 1>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)dSummed.data_ptr(),
                         feats.data_ptr<int64_t>(), dPre.data_ptr(),
                         dBias32.data_ptr(), dPad32.data_ptr(),
                         S, nF, HP, pad_row);
    else if (mode == 2)
      hipLaunchKernelGGL((dpre_scatter_kernel<scalar_t, 2>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)dSummed.data_ptr(),
                         feats.data_ptr<int64_t>(), dPre.data_ptr(),
                         dBias32.data_ptr(), dPad32.data_ptr(),
                         S, nF, HP, pad_row);
    else
      hipLaunchKernelGGL((dpre_scatter_kernel<scalar_t, 0>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)dSummed.data_ptr(),
                         feats.data_ptr<int64_t>(), dPre.data_ptr(),
                         dBias32.data_ptr(), dPad32.data_ptr(),
                         S, nF, HP, pad_row);
  });
  return {finish(dBias32), finish(dPad32)};
}

// Doc-major atomic-free variant (GPU-state-machine arenas): one block per
// doc, LDS accumulation, plain stores into an UNINITIALIZED dPre — see
// dpre_docmajor_kernel.  Same return contract as dpre_scatter.
std::vector<at::Tensor> dpre_scatter_docmajor(at::Tensor dSummed,
                                              at::Tensor feats, at::Tensor dPre,
                                              at::Tensor off, at::Tensor lens,
                                              int64_t pad_row, int64_t cap_mult,
                                              int64_t maxlen) {
  check_dev(dSummed);
  TORCH_CHECK(feats.scalar_type() == at::kLong);
  TORCH_CHECK(dPre.scalar_type() == dSummed.scalar_type(),
              "docmajor dPre accumulates in the compute dtype");
  int nF = (int)feats.size(1);
  int HP = (int)dPre.size(-1);
  long n_docs = off.size(0);
  TORCH_CHECK(HP % SRX_WAVE == 0 && HP <= 128, "docmajor needs HP in {64,128}");
  TORCH_CHECK(nF <= 16, "docmajor nF <= 16");
  size_t lds = (size_t)maxlen * HP * sizeof(float);
  TORCH_CHECK(lds <= 64 * 1024, "docmajor maxlen*HP too large for LDS");
  auto dBias32 = at::zeros({(long)HP}, dSummed.options().dtype(at::kFloat));
  auto dPad32 =
      at::zeros({(long)nF, (long)HP}, dSummed.options().dtype(at::kFloat));
  if (n_docs == 0) return {dBias32, dPad32};
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long>(n_docs, 65535);
  DISPATCH_F(dSummed.scalar_type(), {
    hipLaunchKernelGGL((dpre_docmajor_kernel<scalar_t>),
                       dim3(grid, (unsigned)nF), dim3(256), lds, stream,
                       (const scalar_t*)dSummed.data_ptr(),
                       feats.data_ptr<int64_t>(),
                       (scalar_t*)dPre.data_ptr(),
                       dBias32.data_ptr<float>(), dPad32.data_ptr<float>(),
                       off.data_ptr<int32_t>(), lens.data_ptr<int32_t>(),
                       n_docs, pad_row, nF, HP, (int)cap_mult);
  });
  return {dBias32, dPad32};
}

// ----------------------------------------------------- activations
// Thinc's elementwise activation surface (mish/swish/gelu/clipped_linear
// — srx_activations.hip.h); OP codes shared with ops/api.py.
template <int OP>
static void act_launch(bool bwd, const at::Tensor& A, const at::Tensor& B,
                       at::Tensor& out, double slope, double offset, double lo,
                       double hi) {
  long n = out.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(out.scalar_type(), {
    constexpr int V = sizeof(scalar_t) == 2 ? 8 : 4;
    if (n % V == 0) {
      long chunks = n / V;
      if (bwd)
        hipLaunchKernelGGL((act_bwd_kernel<scalar_t, V, OP>),
                           dim3(grid_for(chunks)), dim3(kBlock), 0, stream,
                           (const scalar_t*)A.data_ptr(),
                           (const scalar_t*)B.data_ptr(),
                           (scalar_t*)out.data_ptr(), chunks, (float)slope,
                           (float)offset, (float)lo, (float)hi);
      else
        hipLaunchKernelGGL((act_fwd_kernel<scalar_t, V, OP>),
                           dim3(grid_for(chunks)), dim3(kBlock), 0, stream,
                           (const scalar_t*)A.data_ptr(),
                           (scalar_t*)out.data_ptr(), chunks, (float)slope,
                           (float)offset, (float)lo, (float)hi);
    } else {
      if (bwd)
        hipLaunchKernelGGL((act_bwd_kernel<scalar_t, 1, OP>),
                           dim3(grid_for(n)), dim3(kBlock), 0, stream,
                           (const scalar_t*)A.data_ptr(),
                           (const scalar_t*)B.data_ptr(),
                           (scalar_t*)out.data_ptr(), n, (float)slope,
                           (float)offset, (float)lo, (float)hi);
      else
        hipLaunchKernelGGL((act_fwd_kernel<scalar_t, 1, OP>),
                           dim3(grid_for(n)), dim3(kBlock), 0, stream,
                           (const scalar_t*)A.data_ptr(),
                           (scalar_t*)out.data_ptr(), n, (float)slope,
                           (float)offset, (float)lo, (float)hi);
    }
  });
}

at::Tensor act_fwd(at::Tensor X, int64_t op, double slope, double offset,
                   double lo, double hi) {
  check_dev(X);
  auto Xc = X.contiguous();
  auto Y = at::empty_like(Xc);
  if (Y.numel() == 0) return Y;
  at::Tensor dummy;
  switch (op) {
    case SRX_ACT_MISH: act_launch<SRX_ACT_MISH>(false, Xc, dummy, Y, slope, offset, lo, hi); break;
    case SRX_ACT_SWISH: act_launch<SRX_ACT_SWISH>(false, Xc, dummy, Y, slope, offset, lo, hi); break;
    case SRX_ACT_GELU: act_launch<SRX_ACT_GELU>(false, Xc, dummy, Y, slope, offset, lo, hi); break;
    default: act_launch<SRX_ACT_CLIPPED_LINEAR>(false, Xc, dummy, Y, slope, offset, lo, hi); break;
  }
  return Y;
}

at::Tensor act_bwd(at::Tensor dY, at::Tensor X, int64_t op, double slope,
                   double offset, double lo, double hi) {
  check_dev(dY);
  auto dYc = dY.contiguous();
  auto Xc = X.contiguous();
  auto dX = at::empty_like(Xc);
  if (dX.numel() == 0) return dX;
  switch (op) {
    case SRX_ACT_MISH: act_launch<SRX_ACT_MISH>(true, dYc, Xc, dX, slope, offset, lo, hi); break;
    case SRX_ACT_SWISH: act_launch<SRX_ACT_SWISH>(true, dYc, Xc, dX, slope, offset, lo, hi); break;
    case SRX_ACT_GELU: act_launch<SRX_ACT_GELU>(true, dYc, Xc, dX, slope, offset, lo, hi); break;
    default: act_launch<SRX_ACT_CLIPPED_LINEAR>(true, dYc, Xc, dX, slope, offset, lo, hi); break;
  }
  return dX;
}

// ----------------------------------------------------- attention softmax
// Fused masked softmax (+dropout) between the two attention bmm GEMMs —
// srx_attn.hip.h.  S/P/dS: [NH, L, L]; lens: [NH/heads] int32.
std::vector<at::Tensor> attn_softmax_fwd(at::Tensor S, at::Tensor lens,
                                         int64_t heads, double scale,
                                         double drop_p, int64_t seed) {
  check_dev(S);
  long NH = S.size(0);
  int L = (int)S.size(-1);
  TORCH_CHECK(L <= SRX_ATTN_MAX_L, "attn softmax: L <= 256");
  auto P = at::empty_like(S);
  auto lse = at::empty({NH, (long)L}, S.options().dtype(at::kFloat));
  long n_rows = NH * L;
  if (n_rows == 0) return {P, lse};
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(n_rows * SRX_WAVE);
  float keep = 1.0f - (float)drop_p;
  DISPATCH_F(S.scalar_type(), {
    if (drop_p > 0.0)
      hipLaunchKernelGGL((attn_softmax_fwd_kernel<scalar_t, true>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)S.data_ptr(), (scalar_t*)P.data_ptr(),
                         lse.data_ptr<float>(), lens.data_ptr<int32_t>(),
                         n_rows, L, (int)heads, (float)scale, keep,
                         (unsigned long long)seed);
    else
      hipLaunchKernelGGL((attn_softmax_fwd_kernel<scalar_t, false>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)S.data_ptr(), (scalar_t*)P.data_ptr(),
                         lse.data_ptr<float>(), lens.data_ptr<int32_t>(),
                         n_rows, L, (int)heads, (float)scale, keep,
                         (unsigned long long)seed);
  });
  return {P, lse};
}

at::Tensor attn_softmax_bwd(at::Tensor S, at::Tensor lse, at::Tensor dPt,
                            at::Tensor lens, int64_t heads, double scale,
                            double drop_p, int64_t seed) {
  check_dev(S);
  long NH = S.size(0);
  int L = (int)S.size(-1);
  auto dS = at::empty_like(S);
  long n_rows = NH * L;
  if (n_rows == 0) return dS;
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(n_rows * SRX_WAVE);
  float keep = 1.0f - (float)drop_p;
  DISPATCH_F(S.scalar_type(), {
    if (drop_p > 0.0)
      hipLaunchKernelGGL((attn_softmax_bwd_kernel<scalar_t, true>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)S.data_ptr(), lse.data_ptr<float>(),
                         (const scalar_t*)dPt.data_ptr(),
                         (scalar_t*)dS.data_ptr(), lens.data_ptr<int32_t>(),
                         n_rows, L, (int)heads, (float)scale, keep,
                         (unsigned long long)seed);
    else
      hipLaunchKernelGGL((attn_softmax_bwd_kernel<scalar_t, false>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const scalar_t*)S.data_ptr(), lse.data_ptr<float>(),
                         (const scalar_t*)dPt.data_ptr(),
                         (scalar_t*)dS.data_ptr(), lens.data_ptr<int32_t>(),
                         n_rows, L, (int)heads, (float)scale, keep,
                         (unsigned long long)seed);
  });
  return dS;
}

// ------------------------------------------- fused flash-style attention
std::vector<at::Tensor> attn_fused_fwd(at::Tensor Q, at::Tensor K,
                                       at::Tensor V, at::Tensor lens,
                                       int64_t heads, double scale,
                                       double drop_p, int64_t seed) {
  check_dev(Q);
  TORCH_CHECK(Q.scalar_type() == at::kBFloat16, "fused attention is bf16-only");
  long NH = Q.size(0);
  int L = (int)Q.size(1);
  TORCH_CHECK((int)Q.size(2) == 64, "fused attention needs D == 64");
  TORCH_CHECK(L <= SRX_ATTN_FUSED_MAX_L, "fused attention: L <= 96");
  auto O = at::empty_like(Q);
  auto lse = at::empty({NH, (long)L}, Q.options().dtype(at::kFloat));
  if (NH == 0) return {O, lse};
  int Lp = (L + 31) & ~31;
  int NT = Lp / 32;
  size_t lds =
      ((size_t)2 * Lp * SRX_ATTN_LDQ + 64 * SRX_ATTN_LDT + (size_t)Lp * SRX_ATTN_LDT) * 2;
  static std::once_flag attr_f;
  std::call_once(attr_f, []() {
#define AF_ATTR(DR, NTV)                                                      \
    (void)hipFuncSetAttribute((const void*)attn_fused_fwd_kernel<DR, NTV>,    \
                              hipFuncAttributeMaxDynamicSharedMemorySize,     \
                              100 * 1024)
    AF_ATTR(true, 1); AF_ATTR(true, 2); AF_ATTR(true, 3);
    AF_ATTR(false, 1); AF_ATTR(false, 2); AF_ATTR(false, 3);
#undef AF_ATTR
  });
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long>(NH, 65535);
  float keep = 1.0f - (float)drop_p;
#define LAUNCH_AF(DR, NTV)                                                    \
  hipLaunchKernelGGL((attn_fused_fwd_kernel<DR, NTV>), dim3(grid),            \
                     dim3(64 * NTV), lds, stream,                             \
                     (const bf16_t*)Q.data_ptr(), (const bf16_t*)K.data_ptr(),\
                     (const bf16_t*)V.data_ptr(), lens.data_ptr<int32_t>(),   \
                     (bf16_t*)O.data_ptr(), lse.data_ptr<float>(), NH, L,     \
                     (int)heads, (float)scale, keep, (unsigned long long)seed)
  if (drop_p > 0.0) {
    if (NT == 1) LAUNCH_AF(true, 1); else if (NT == 2) LAUNCH_AF(true, 2); else LAUNCH_AF(true, 3);
  } else {
    if (NT == 1) LAUNCH_AF(false, 1); else if (NT == 2) LAUNCH_AF(false, 2); else LAUNCH_AF(false, 3);
  }
#undef LAUNCH_AF
  return {O, lse};
}

std::vector<at::Tensor> attn_fused_bwd(at::Tensor Q, at::Tensor K,
                                       at::Tensor V, at::Tensor dO,
                                       at::Tensor lse, at::Tensor lens,
                                       int64_t heads, double scale,
                                       double drop_p, int64_t seed) {
  check_dev(Q);
  long NH = Q.size(0);
  int L = (int)Q.size(1);
  auto dQ = at::empty_like(Q);
  auto dK = at::empty_like(Q);
  auto dV = at::empty_like(Q);
  if (NH == 0) return {dQ, dK, dV};
  int Lp = (L + 31) & ~31;
  int NT = Lp / 32;
  size_t lds = ((size_t)4 * Lp * SRX_ATTN_LDQ + (size_t)3 * 64 * SRX_ATTN_LDT +
                (size_t)2 * Lp * SRX_ATTN_LDT) * 2;
  static std::once_flag attr_b;
  std::call_once(attr_b, []() {
#define AB_ATTR(DR, NTV)                                                      \
    (void)hipFuncSetAttribute((const void*)attn_fused_bwd_kernel<DR, NTV>,    \
                              hipFuncAttributeMaxDynamicSharedMemorySize,     \
                              144 * 1024)
    AB_ATTR(true, 1); AB_ATTR(true, 2); AB_ATTR(true, 3);
    AB_ATTR(false, 1); AB_ATTR(false, 2); AB_ATTR(false, 3);
#undef AB_ATTR
  });
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long>(NH, 65535);
  float keep = 1.0f - (float)drop_p;
#define LAUNCH_AB(DR, NTV)                                                    \
  hipLaunchKernelGGL((attn_fused_bwd_kernel<DR, NTV>), dim3(grid),            \
                     dim3(64 * NTV), lds, stream,                             \
                     (const bf16_t*)Q.data_ptr(), (const bf16_t*)K.data_ptr(),\
                     (const bf16_t*)V.data_ptr(), (const bf16_t*)dO.data_ptr(),\
                     lse.data_ptr<float>(), lens.data_ptr<int32_t>(),         \
                     (bf16_t*)dQ.data_ptr(), (bf16_t*)dK.data_ptr(),          \
                     (bf16_t*)dV.data_ptr(), NH, L, (int)heads,               \
                     (float)scale, keep, (unsigned long long)seed)
  if (drop_p > 0.0) {
    if (NT == 1) LAUNCH_AB(true, 1); else if (NT == 2) LAUNCH_AB(true, 2); else LAUNCH_AB(true, 3);
  } else {
    if (NT == 1) LAUNCH_AB(false, 1); else if (NT == 2) LAUNCH_AB(false, 2); else LAUNCH_AB(false, 3);
  }
#undef LAUNCH_AB
  return {dQ, dK, dV};
}

// ----------------------------------------------------- dropout mask
at::Tensor dropout_mask(at::Tensor like, double p, int64_t seed, int64_t offset) {
  check_dev(like);
  auto out = at::empty_like(like);
  long n = out.numel();
  if (n == 0 || p <= 0.0) {
    out.fill_(1.0);
    return out;
  }
  float keep = 1.0f - (float)p;
  float scale = 1.0f / keep;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(like.scalar_type(), {
    hipLaunchKernelGGL((dropout_mask_kernel<scalar_t>), dim3(grid_for(n, 4)),
                       dim3(kBlock), 0, stream, (scalar_t*)out.data_ptr(), n,
                       keep, scale, (unsigned long long)seed,
                       (unsigned long long)offset);
  });
  return out;
}

// ------------------------------------------------------- softmax + CE
// returns (loss_and_count fp32 [2], dScores) — dScores = softmax - onehot
// (unnormalized; the python wrapper divides by the valid count).
std::vector<at::Tensor> softmax_ce(at::Tensor scores, at::Tensor gold) {
  check_dev(scores);
  TORCH_CHECK(gold.scalar_type() == at::kLong);
  long N = scores.size(0);
  int C = (int)scores.size(1);
  auto dScores = at::empty_like(scores);
  auto loss = at::zeros({2}, scores.options().dtype(at::kFloat));
  if (N == 0) return {loss, dScores};
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long>((N + 3) / 4, 2048);
  DISPATCH_F(scores.scalar_type(), {
    hipLaunchKernelGGL((softmax_ce_kernel<scalar_t>), dim3(grid), dim3(kBlock), 0,
                       stream, (const scalar_t*)scores.data_ptr(),
                       gold.data_ptr<int64_t>(), (scalar_t*)dScores.data_ptr(),
                       loss.data_ptr<float>(), N, C);
  });
  return {loss, dScores};
}

// ---------------------------------------------- segmented reductions
std::vector<at::Tensor> reduce_ragged(at::Tensor X, at::Tensor offsets, int64_t mode) {
  check_dev(X);
  long N = offsets.size(0) - 1;
  int W = (int)X.size(1);
  auto out = at::empty({N, (long)W}, X.options());
  auto argmax = mode == 2 ? at::empty({N, (long)W}, X.options().dtype(at::kInt))
                          : at::empty({0}, X.options().dtype(at::kInt));
  if (N == 0) return {out, argmax};
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(N * SRX_WAVE);
  DISPATCH_F(X.scalar_type(), {
    if (mode == 0)
      hipLaunchKernelGGL((reduce_ragged_kernel<scalar_t, 0>), dim3(grid), dim3(kBlock),
                         0, stream, (const scalar_t*)X.data_ptr(),
                         offsets.data_ptr<int32_t>(), (scalar_t*)out.data_ptr(),
                         nullptr, N, W);
    else if (mode == 1)
      hipLaunchKernelGGL((reduce_ragged_kernel<scalar_t, 1>), dim3(grid), dim3(kBlock),
                         0, stream, (const scalar_t*)X.data_ptr(),
                         offsets.data_ptr<int32_t>(), (scalar_t*)out.data_ptr(),
                         nullptr, N, W);
    else
      hipLaunchKernelGGL((reduce_ragged_kernel<scalar_t, 2>), dim3(grid), dim3(kBlock),
                         0, stream, (const scalar_t*)X.data_ptr(),
                         offsets.data_ptr<int32_t>(), (scalar_t*)out.data_ptr(),
                         argmax.data_ptr<int32_t>(), N, W);
  });
  return {out, argmax};
}

at::Tensor reduce_ragged_bwd(at::Tensor dY, at::Tensor doc_of, at::Tensor offsets,
                             int64_t Ttot, int64_t mode) {
  check_dev(dY);
  int W = (int)dY.size(1);
  auto dX = at::empty({Ttot, (long)W}, dY.options());
  if (Ttot == 0) return dX;
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(dY.scalar_type(), {
    if (mode == 0)
      hipLaunchKernelGGL((reduce_ragged_bwd_kernel<scalar_t, 0>),
                         dim3(grid_for(Ttot * (long)W)), dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), doc_of.data_ptr<int32_t>(),
                         offsets.data_ptr<int32_t>(), (scalar_t*)dX.data_ptr(), Ttot, W);
    else
      hipLaunchKernelGGL((reduce_ragged_bwd_kernel<scalar_t, 1>),
                         dim3(grid_for(Ttot * (long)W)), dim3(kBlock), 0, stream,
                         (const scalar_t*)dY.data_ptr(), doc_of.data_ptr<int32_t>(),
                         offsets.data_ptr<int32_t>(), (scalar_t*)dX.data_ptr(), Ttot, W);
  });
  return dX;
}

at::Tensor reduce_max_bwd(at::Tensor dY, at::Tensor argmax, int64_t Ttot) {
  check_dev(dY);
  long N = dY.size(0);
  int W = (int)dY.size(1);
  auto dX = at::zeros({Ttot, (long)W}, dY.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  DISPATCH_F(dY.scalar_type(), {
    hipLaunchKernelGGL((reduce_max_bwd_kernel<scalar_t>), dim3(grid_for(N * (long)W)),
                       dim3(kBlock), 0, stream, (const scalar_t*)dY.data_ptr(),
                       argmax.data_ptr<int32_t>(), (scalar_t*)dX.data_ptr(), N, W);
  });
  return dX;
}

// ----------------------------------------- fused MWE backward stage 1
// (dropmask x dY -> LN bwd -> maxout scatter) + dg/db/dbias column sums in
// ONE kernel; see mwe_bwd_stage1_kernel.
std::vector<at::Tensor> mwe_bwd_stage1(at::Tensor dY,
                                       c10::optional<at::Tensor> dropmask,
                                       at::Tensor Mout, at::Tensor g,
                                       at::Tensor mu, at::Tensor rstd,
                                       at::Tensor which,
                                       bool deterministic = false) {
  check_dev(dY);
  long N = dY.size(0);
  int W = (int)dY.size(1);
  TORCH_CHECK(W <= 256, "mwe_bwd_stage1 W <= 256");
  auto dPre = at::empty({N, 3L * W}, dY.options());
  auto acc_dt = deterministic ? at::kLong : at::kFloat;
  auto dg32 = at::zeros({W}, dY.options().dtype(acc_dt));
  auto db32 = at::zeros({W}, dY.options().dtype(acc_dt));
  auto dbias32 = at::zeros({3L * W}, dY.options().dtype(acc_dt));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (N > 0) {
    int grid = (int)std::min<long>((N + 3) / 4, 4096);
    DISPATCH_F(dY.scalar_type(), {
      const scalar_t* mask =
          dropmask ? (const scalar_t*)dropmask->data_ptr() : nullptr;
      if (deterministic)
        hipLaunchKernelGGL((mwe_bwd_stage1_kernel<scalar_t, true>), dim3(grid),
                           dim3(kBlock), 0, stream,
                           (const scalar_t*)dY.data_ptr(), mask,
                           (const scalar_t*)Mout.data_ptr(),
                           (const scalar_t*)g.data_ptr(), mu.data_ptr<float>(),
                           rstd.data_ptr<float>(), which.data_ptr<uint8_t>(),
                           (scalar_t*)dPre.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), dbias32.data_ptr(), N, W);
      else
        hipLaunchKernelGGL((mwe_bwd_stage1_kernel<scalar_t, false>), dim3(grid),
                           dim3(kBlock), 0, stream,
                           (const scalar_t*)dY.data_ptr(), mask,
                           (const scalar_t*)Mout.data_ptr(),
                           (const scalar_t*)g.data_ptr(), mu.data_ptr<float>(),
                           rstd.data_ptr<float>(), which.data_ptr<uint8_t>(),
                           (scalar_t*)dPre.data_ptr(), dg32.data_ptr(),
                           db32.data_ptr(), dbias32.data_ptr(), N, W);
    });
  }
  if (deterministic) {
    double s = 1.0 / 16777216.0;
    return {dPre, (dg32.to(at::kFloat) * s), (db32.to(at::kFloat) * s),
            (dbias32.to(at::kFloat) * s)};
  }
  return {dPre, dg32, db32, dbias32};
}

// ------------------------------------------------ fused MWE layer (MFMA)
template <int W>
void launch_mwe(const at::Tensor& X, const at::Tensor& Wt, const at::Tensor& bias,
                const at::Tensor& g, const at::Tensor& b, const at::Tensor& starts,
                const at::Tensor& ends, const c10::optional<at::Tensor>& dropmask,
                at::Tensor& Y, at::Tensor& Mout,
                at::Tensor& which, at::Tensor& mu, at::Tensor& rstd, long T,
                double eps, hipStream_t stream) {
  constexpr int WP = W + 8;
  constexpr int NW = W / 32;
  // per-section staging: [64][WP] A + [3W][WP] B (73 KB at W=96 -> two
  // blocks per CU; the old all-sections layout was 101 KB -> one block)
  size_t lds_bytes = (size_t)(64 * WP + 3 * W * WP) * sizeof(bf16_t) +
                     (size_t)(2 * NW * 64) * sizeof(float);
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute((const void*)mwe_layer_fwd_kernel<W>,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)lds_bytes);
    attr_set = true;
  }
  dim3 grid((unsigned)(T / 64));
  hipLaunchKernelGGL((mwe_layer_fwd_kernel<W>), grid, dim3(64 * NW), lds_bytes,
                     stream, (const bf16_t*)X.data_ptr(), (const bf16_t*)Wt.data_ptr(),
                     (const bf16_t*)bias.data_ptr(), (const bf16_t*)g.data_ptr(),
                     (const bf16_t*)b.data_ptr(), starts.data_ptr<uint8_t>(),
                     ends.data_ptr<uint8_t>(),
                     dropmask ? (const bf16_t*)dropmask->data_ptr() : nullptr,
                     (bf16_t*)Y.data_ptr(),
                     (bf16_t*)Mout.data_ptr(), which.data_ptr<uint8_t>(),
                     mu.data_ptr<float>(), rstd.data_ptr<float>(), T, (float)eps);
}

std::vector<at::Tensor> mwe_layer_fwd(at::Tensor X, at::Tensor Wt, at::Tensor bias,
                                      at::Tensor g, at::Tensor b, at::Tensor starts,
                                      at::Tensor ends,
                                      c10::optional<at::Tensor> dropmask, double eps) {
  check_dev(X);
  TORCH_CHECK(X.scalar_type() == at::kBFloat16, "mwe_layer is bf16-only");
  long T = X.size(0);
  int W = (int)X.size(1);
  TORCH_CHECK(W == 96 || W == 128, "mwe_layer supports W in {96,128}");
  TORCH_CHECK(T % 64 == 0, "mwe_layer needs T % 64 == 0 (TokenBatch pads)");
  TORCH_CHECK(Wt.size(0) == 3 * W && Wt.size(1) == 3 * W, "weight must be [3W,3W]");
  auto Y = at::empty_like(X);
  auto Mout = at::empty_like(X);
  auto which = at::empty({T, (long)W}, X.options().dtype(at::kByte));
  auto mu = at::empty({T}, X.options().dtype(at::kFloat));
  auto rstd = at::empty({T}, X.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (T > 0) {
    if (W == 96)
      launch_mwe<96>(X, Wt, bias, g, b, starts, ends, dropmask, Y, Mout, which, mu,
                     rstd, T, eps, stream);
    else
      launch_mwe<128>(X, Wt, bias, g, b, starts, ends, dropmask, Y, Mout, which, mu,
                      rstd, T, eps, stream);
  }
  return {Y, Mout, which, mu, rstd};
}

// --------------------------------------------- fused transition step (C++)
// One python call per transition step: state scorer kernel + upper GEMM +
// on-device action selection, with a C++ autograd node so the per-step
// python op count collapses (~10 -> 1).  Backward stashes the compact
// (feats, dSummed) pair into a session store keyed by task id; the python
// side drains it for the batched sorted dPre scatter.
// NOTE: backward runs on autograd worker threads WITHOUT the GIL — only
// ATen ops and the mutex-guarded store are touched there.
namespace fusedstep {

struct Store {
  std::mutex mu;
  std::unordered_map<int64_t, std::vector<std::pair<at::Tensor, at::Tensor>>> m;
};
Store& store() {
  static Store s;
  return s;
}

class Fn : public torch::autograd::Function<Fn> {
 public:
  static torch::autograd::variable_list forward(
      torch::autograd::AutogradContext* ctx, at::Tensor pre_d, at::Tensor feats,
      at::Tensor lower_b, at::Tensor upperW, at::Tensor upperB,
      at::Tensor is_gold, at::Tensor valid, int64_t task_id, bool train) {
    auto hw = parser_step_fwd(pre_d, feats, lower_b);
    auto hidden = hw[0];
    auto which = hw[1];
    auto scores = at::addmm(upperB, hidden, upperW.t());
    auto actions = action_select(scores, is_gold, valid);
    ctx->save_for_backward({feats, which, hidden, upperW});
    ctx->saved_data["task_id"] = task_id;
    ctx->saved_data["train"] = train;
    ctx->mark_non_differentiable({actions});
    return {scores, actions};
  }

  static torch::autograd::variable_list backward(
      torch::autograd::AutogradContext* ctx, torch::autograd::variable_list grads) {
    auto saved = ctx->get_saved_variables();
    auto feats = saved[0];
    auto which = saved[1];
    auto hidden = saved[2];
    auto upperW = saved[3];
    auto dScores = grads[0].contiguous();
    auto dUpperW = dScores.t().mm(hidden);
    auto dUpperB = dScores.sum(0);
    auto dHidden = dScores.mm(upperW).contiguous();
    auto dSummed = maxout_bwd(dHidden, which, 2).view({dHidden.size(0), -1});
    auto dLowerB = dSummed.to(at::kFloat).sum(0);  // mirrors the python accum path
    int64_t task_id = ctx->saved_data["task_id"].toInt();
    {
      auto& s = store();
      std::lock_guard<std::mutex> lock(s.mu);
      s.m[task_id].emplace_back(feats, dSummed);
    }
    // pre_d is detached by contract (its gradient flows via the batched
    // scatter + inject_grad); feats/is_gold/valid are integer masks.
    return {at::Tensor(), at::Tensor(), dLowerB, dUpperW, dUpperB,
            at::Tensor(), at::Tensor(), at::Tensor(), at::Tensor()};
  }
};

std::vector<at::Tensor> fused_step(at::Tensor pre_d, at::Tensor feats,
                                   at::Tensor lower_b, at::Tensor upperW,
                                   at::Tensor upperB, at::Tensor is_gold,
                                   at::Tensor valid, int64_t task_id, bool train) {
  auto out = Fn::apply(pre_d, feats, lower_b, upperW, upperB, is_gold, valid,
                       task_id, train);
  return {out[0], out[1]};
}

std::vector<std::vector<at::Tensor>> fused_entries_take(int64_t task_id) {
  auto& s = store();
  std::lock_guard<std::mutex> lock(s.mu);
  std::vector<std::vector<at::Tensor>> out;
  auto it = s.m.find(task_id);
  if (it != s.m.end()) {
    for (auto& pr : it->second) out.push_back({pr.first, pr.second});
    s.m.erase(it);
  }
  return out;
}

}  // namespace fusedstep

}  // namespace

// srx_steploop.hip — the C++-owned transition loop (one python call/batch)
std::vector<std::vector<at::Tensor>> srx_run_transition_loop(
    std::vector<std::tuple<int64_t, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, bool>>
        tasks);

// srx_gpustate.hip — fully GPU-resident state machines (one wave per doc)
std::vector<at::Tensor> srx_gpu_arceager(
    at::Tensor pre, at::Tensor off, at::Tensor lens, at::Tensor gh,
    at::Tensor gl, at::Tensor kids_off, at::Tensor kids, at::Tensor lowerB,
    at::Tensor upperW, at::Tensor upperB, int64_t total, int64_t n_labels,
    bool train);
std::vector<at::Tensor> srx_gpu_biluo(
    at::Tensor pre, at::Tensor off, at::Tensor lens, at::Tensor gold,
    at::Tensor lowerB, at::Tensor upperW, at::Tensor upperB, int64_t total,
    int64_t n_types, bool train);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("seq2col_fwd", &seq2col_fwd);
  m.def("seq2col_bwd", &seq2col_bwd, py::arg("dY"), py::arg("starts"),
        py::arg("ends"), py::arg("residual") = py::none());
  m.def("mwe_bwd_stage1", &mwe_bwd_stage1, py::arg("dY"), py::arg("dropmask"),
        py::arg("Mout"), py::arg("g"), py::arg("mu"), py::arg("rstd"),
        py::arg("which"), py::arg("deterministic") = false);
  m.def("maxout_fwd", &maxout_fwd);
  m.def("maxout_bwd", &maxout_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd, py::arg("dY"), py::arg("X"),
        py::arg("g"), py::arg("mu"), py::arg("rstd"),
        py::arg("deterministic") = false);
  m.def("hashembed_fwd", &hashembed_fwd, py::arg("table"), py::arg("ids"),
        py::arg("seed"), py::arg("out") = py::none(), py::arg("col_off") = 0);
  m.def("hashembed_bwd", &hashembed_bwd);
  m.def("parser_step_fwd", &parser_step_fwd);
  m.def("parser_step_bwd", &parser_step_bwd);
  m.def("parser_step_bwd_into", &parser_step_bwd_into);
  m.def("action_select", &action_select);
  m.def("seg_scatter_add", &seg_scatter_add);
  m.def("adam_step", &adam_step);
  m.def("dropout_mask", &dropout_mask);
  m.def("act_fwd", &act_fwd);
  m.def("attn_softmax_fwd", &attn_softmax_fwd);
  m.def("attn_fused_fwd", &attn_fused_fwd);
  m.def("attn_fused_bwd", &attn_fused_bwd);
  m.def("attn_softmax_bwd", &attn_softmax_bwd);
  m.def("act_bwd", &act_bwd);
  m.def("softmax_ce", &softmax_ce);
  m.def("reduce_ragged", &reduce_ragged);
  m.def("reduce_ragged_bwd", &reduce_ragged_bwd);
  m.def("reduce_max_bwd", &reduce_max_bwd);
  m.def("mwe_layer_fwd", &mwe_layer_fwd);
  m.def("fused_step", &fusedstep::fused_step);
  m.def("fused_entries_take", &fusedstep::fused_entries_take);
  m.def("transition_ce", &transition_ce, py::arg("scores"), py::arg("gold"),
        py::arg("valid"), py::arg("deterministic") = false);
  m.attr("FIXED_SCALE") = 16777216.0;
  m.def("dpre_scatter", &dpre_scatter);
  m.def("dpre_scatter_docmajor", &dpre_scatter_docmajor);
  m.def("run_transition_loop", &srx_run_transition_loop,
        py::call_guard<py::gil_scoped_release>());
  m.def("gpu_arceager", &srx_gpu_arceager, py::arg("pre"), py::arg("off"),
        py::arg("lens"), py::arg("gh"), py::arg("gl"), py::arg("kids_off"),
        py::arg("kids"), py::arg("lowerB"), py::arg("upperW"),
        py::arg("upperB"), py::arg("total"), py::arg("n_labels"),
        py::arg("train"));
  m.def("gpu_biluo", &srx_gpu_biluo, py::arg("pre"), py::arg("off"),
        py::arg("lens"), py::arg("gold"), py::arg("lowerB"), py::arg("upperW"),
        py::arg("upperB"), py::arg("total"), py::arg("n_types"),
        py::arg("train"));
  m.attr("GPU_STATE_MAXLEN") = 128;
}
