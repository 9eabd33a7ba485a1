"""ZeRO-1 engine: flat-buffer data parallelism with sharded Adam.

The MI355X re-architecture of the reference's parameter-sharding scheme
(SURVEY.md §2.3): where spacy-ray partitions Thinc param keys across Ray
actors and pushes version-stamped grads/params asynchronously
(`/root/reference/spacy_ray/proxies.py:9-133`, `util.py:57-75`), this engine
keeps the same math — each rank owns a shard of the parameters and runs the
optimizer only on it — expressed synchronously as collectives over xGMI:

  backward  -> per-bucket gradient reduce-scatter (RCCL) launched from
               post-accumulate-grad hooks on a side HIP stream, overlapping
               the rest of backward;
  step      -> global-norm clip (one scalar all-reduce) + Adam on the
               rank's contiguous fp32 master shard (bf16 params on GPU);
  publish   -> per-bucket parameter all-gather.

Layout: all trainable params live in ONE flat device buffer, partitioned
into buckets of ~bucket_bytes at parameter boundaries; each bucket is padded
so `world` divides it, and rank r owns slice r of every bucket.  Optimizer
state (master/m/v) is one contiguous fp32 tensor per kind covering the
rank's slices — so the whole Adam step is a handful of elementwise kernels
over single contiguous tensors (one fused HIP kernel in ops/kernels).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from spacy_ray_amd.train.optimizer import AdamSpec
from .comm import Comm, LocalComm


class _Bucket:
    __slots__ = ("index", "start", "end", "per", "shard_off", "params", "ready", "launched")

    def __init__(self, index: int, start: int) -> None:
        self.index = index
        self.start = start
        self.end = start
        self.per = 0
        self.shard_off = 0
        self.params: List[torch.nn.Parameter] = []
        self.ready = 0
        self.launched = False


class ZeRO1Engine:
    def __init__(
        self,
        nlp,
        spec: AdamSpec,
        comm: Optional[Comm] = None,
        *,
        bucket_bytes: int = 16 << 20,
        overlap: bool = True,
        compute_dtype: Optional[torch.dtype] = None,
    ) -> None:
        self.nlp = nlp
        self.spec = spec
        self.comm = comm or LocalComm()
        self.device = nlp.device
        self.is_cuda = self.device.type == "cuda"
        if compute_dtype is None:
            compute_dtype = torch.bfloat16 if self.is_cuda else torch.float32
        self.dtype = compute_dtype
        self.overlap = overlap and self.is_cuda
        self.module = nlp.torch_module()
        self.step_count = 0
        self._sync = False

        world = self.comm.world
        align = 64 * world
        named = [(n, p) for n, p in self.module.named_parameters() if p.requires_grad]
        self.param_names = [n for n, _ in named]
        params = [p for _, p in named]

        # ---- bucket construction at parameter boundaries
        # Every param starts at a 64-element boundary: an odd element offset
        # gives hipBLASLt a 2-byte-aligned weight pointer and it drops to a
        # slow path (measured 16 ms for a 116-GFLOP GEMM vs 0.3 ms aligned).
        self.buckets: List[_Bucket] = []
        b = _Bucket(0, 0)
        offset = 0
        self._offsets: List[int] = []
        for p in params:
            offset = -(-offset // 64) * 64
            n = p.numel()
            self._offsets.append(offset)
            b.params.append(p)
            offset += n
            if (offset - b.start) * self.dtype.itemsize >= bucket_bytes:
                offset = -(-offset // align) * align  # pad bucket end
                b.end = offset
                self.buckets.append(b)
                b = _Bucket(len(self.buckets), offset)
        if b.params:
            offset = -(-offset // align) * align
            b.end = offset
            self.buckets.append(b)
        total = offset
        shard_off = 0
        for b in self.buckets:
            b.per = (b.end - b.start) // world
            b.shard_off = shard_off
            shard_off += b.per
        self.shard_elems = shard_off

        # ---- flat buffers
        self.flat_param = torch.zeros(total, dtype=self.dtype, device=self.device)
        self.flat_grad = torch.zeros(total, dtype=self.dtype, device=self.device)
        for p, off in zip(params, self._offsets):
            self.flat_param[off : off + p.numel()].copy_(p.data.reshape(-1).to(self.dtype))
            p.data = self.flat_param[off : off + p.numel()].view(p.shape)
            p.grad = self.flat_grad[off : off + p.numel()].view(p.shape)

        # ---- sharded optimizer state (fp32)
        r = self.comm.rank
        self.master = torch.zeros(self.shard_elems, dtype=torch.float32, device=self.device)
        for b in self.buckets:
            src = self.flat_param[b.start + r * b.per : b.start + (r + 1) * b.per]
            self.master[b.shard_off : b.shard_off + b.per].copy_(src.float())
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        self.grad_shard = torch.zeros(self.shard_elems, dtype=self.dtype, device=self.device)
        self._g32 = torch.zeros_like(self.master)
        self._param_shard: Optional[torch.Tensor] = None
        # running parameter average (thinc Adam use_averages contract):
        # updated per step, swapped in around evaluation via averaged_params()
        self.avg: Optional[torch.Tensor] = (
            self.master.clone() if getattr(spec, "use_averages", False) else None
        )

        # ---- overlap machinery
        self.comm_stream = torch.cuda.Stream() if self.is_cuda else None
        self._next_bucket = len(self.buckets) - 1
        self._param_bucket: Dict[int, _Bucket] = {}
        for bkt in self.buckets:
            for p in bkt.params:
                self._param_bucket[id(p)] = bkt
        for p in params:
            p.register_post_accumulate_grad_hook(self._grad_ready_hook)

    # ------------------------------------------------------------ internals
    @staticmethod
    def _hip_ext():
        from spacy_ray_amd.ops.api import hip_ext

        return hip_ext()

    def _grad_ready_hook(self, p: torch.nn.Parameter) -> None:
        if not (self._sync and self.overlap):
            return
        bkt = self._param_bucket[id(p)]
        bkt.ready += 1
        self._drain_ready()

    def _drain_ready(self) -> None:
        """Launch ready buckets in FIXED descending index order.

        Collective order must be identical on every rank; grad-READY order is
        not (e.g. a rank whose batch has no entities never backprops the NER
        head, so that bucket only becomes ready at apply_step).  Draining in
        a fixed order, gated on readiness, keeps NCCL ordering rank-uniform.
        Descending because backward produces grads roughly from the last
        pipeline params (parser/NER heads, phase-1) back to the embeddings —
        so the fixed order still overlaps with backward."""
        while self._next_bucket >= 0:
            bkt = self.buckets[self._next_bucket]
            if bkt.ready < len(bkt.params):
                return
            if not bkt.launched:
                self._launch_bucket(bkt)
            self._next_bucket -= 1

    def _launch_bucket(self, bkt: _Bucket) -> None:
        bkt.launched = True
        seg = self.flat_grad[bkt.start : bkt.end]
        out = self.grad_shard[bkt.shard_off : bkt.shard_off + bkt.per]
        if self.is_cuda:
            ev = torch.cuda.Event()
            ev.record()
            with torch.cuda.stream(self.comm_stream):
                self.comm_stream.wait_event(ev)
                self.comm.reduce_scatter_flat(seg, out)
        else:
            self.comm.reduce_scatter_flat(seg, out)

    def _gather_bucket(self, bkt: _Bucket) -> None:
        r = self.comm.rank
        shard = self.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
        self.comm.all_gather_flat(self.flat_param[bkt.start : bkt.end], shard)

    # ------------------------------------------------------------- stepper
    def accumulate(self, examples, drop: float = 0.0, losses: Optional[Dict] = None,
                   sync: bool = True, token_batch=None) -> None:
        import time as _time

        from spacy_ray_amd.utils import timing

        t0 = _time.perf_counter()
        self._sync = sync
        total, _ = self.nlp.forward_loss(examples, losses=losses, drop=drop,
                                         token_batch=token_batch)
        with timing.phase("bwd/main"):
            if total.requires_grad:
                total.backward()
            # else: every loss-producing pipe is frozen — a valid (if odd)
            # configuration; the step is a no-op rather than a crash
        self._sync = False
        if losses is not None:
            # display losses arrive as detached 0-dim device tensors (pipes
            # defer the .item() so the forward path never host-syncs); ONE
            # conversion point here, after backward is queued
            with timing.phase("bwd/loss_sync"):
                for k, v in losses.items():
                    if torch.is_tensor(v):
                        losses[k] = float(v)
        self.last_compute_ms = (_time.perf_counter() - t0) * 1000

    last_compute_ms: float = 0.0
    last_comm_ms: float = 0.0

    def apply_step(self) -> None:
        import time as _time

        from spacy_ray_amd.utils import timing

        t0 = _time.perf_counter()
        with timing.phase("comm+opt/apply_step"):
            self._apply_step_inner()
        # wall time of collective wait + sharded Adam + republish (approx:
        # async GPU work not synced unless SRX_TIMING=1)
        self.last_comm_ms = (_time.perf_counter() - t0) * 1000

    def _apply_step_inner(self) -> None:
        # launch any bucket the hooks didn't (grad-less params, overlap off)
        # — same fixed descending order as _drain_ready
        for bkt in reversed(self.buckets):
            if not bkt.launched:
                self._launch_bucket(bkt)
        self._next_bucket = len(self.buckets) - 1
        if self.is_cuda:
            torch.cuda.current_stream().wait_stream(self.comm_stream)

        s = self.spec
        lr = s.lr(self.step_count)
        t = self.step_count + 1
        bc1 = 1 - s.beta1 ** t
        bc2 = 1 - s.beta2 ** t
        wd = s.L2 if (s.L2 and s.L2_is_weight_decay) else 0.0
        # global-norm clip: one scalar all-reduce over shard norms
        if s.grad_clip:
            sq = self.grad_shard.float().pow(2).sum()
            self.comm.all_reduce_(sq)
            scale_t = torch.clamp(s.grad_clip / (sq.sqrt() + 1e-12), max=1.0)
        else:
            scale_t = None
        hip = self._hip_ext() if self.is_cuda else None
        if hip is not None and (s.L2_is_weight_decay or not s.L2):
            # ONE fused kernel: clip-scale + decoupled wd + Adam + bf16 cast
            # (SURVEY.md §2.5 fused_adam_sharded); the clip scale stays on
            # device — no host sync in the optimizer step
            if self._param_shard is None:
                self._param_shard = torch.empty_like(self.grad_shard)
            scale_dev = (scale_t.float().reshape(1) if scale_t is not None
                         else torch.empty(0, device=self.device))
            hip.adam_step(self.grad_shard, self.master, self.exp_avg,
                          self.exp_avg_sq, self._param_shard, scale_dev, lr,
                          s.beta1, s.beta2, s.eps, wd, bc1, bc2)
            new_param = self._param_shard
        else:
            self._g32.copy_(self.grad_shard)
            if scale_t is not None:
                self._g32.mul_(scale_t)
            if s.L2 and not s.L2_is_weight_decay:
                self._g32.add_(self.master, alpha=s.L2)
            if wd:
                self.master.mul_(1.0 - lr * wd)
            self.exp_avg.mul_(s.beta1).add_(self._g32, alpha=1 - s.beta1)
            self.exp_avg_sq.mul_(s.beta2).addcmul_(self._g32, self._g32, value=1 - s.beta2)
            denom = (self.exp_avg_sq / bc2).sqrt_().add_(s.eps)
            self.master.addcdiv_(self.exp_avg, denom, value=-lr / bc1)
            new_param = self.master
        # write back + republish
        r = self.comm.rank
        for bkt in self.buckets:
            dst = self.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
            dst.copy_(new_param[bkt.shard_off : bkt.shard_off + bkt.per].to(self.dtype))
            bkt.ready = 0
            bkt.launched = False
        for bkt in self.buckets:
            self._gather_bucket(bkt)
        self.flat_grad.zero_()
        self.grad_shard.zero_()
        self.step_count += 1
        if self.avg is not None:
            # running mean over steps (thinc's averaged weights for eval)
            self.avg.add_(
                (new_param.float() if new_param.dtype != torch.float32 else new_param)
                - self.avg,
                alpha=1.0 / self.step_count,
            )

    def averaged_params(self):
        """Context manager: swap the running parameter AVERAGE into the live
        flat buffer (all-gathered) for evaluation, restore after (thinc
        ``use_averages`` contract).  No-op when use_averages is off."""
        import contextlib

        engine = self

        @contextlib.contextmanager
        def ctx():
            if engine.avg is None:
                yield
                return
            r = engine.comm.rank
            for bkt in engine.buckets:
                dst = engine.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
                dst.copy_(engine.avg[bkt.shard_off : bkt.shard_off + bkt.per].to(engine.dtype))
            for bkt in engine.buckets:
                engine._gather_bucket(bkt)
            try:
                yield
            finally:
                for bkt in engine.buckets:
                    dst = engine.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
                    dst.copy_(engine.master[bkt.shard_off : bkt.shard_off + bkt.per].to(engine.dtype))
                for bkt in engine.buckets:
                    engine._gather_bucket(bkt)

        return ctx()

    # ------------------------------------------------- checkpoint interface
    def refresh_master_from_params(self) -> None:
        """Re-snapshot the fp32 master from the (possibly just-loaded) model
        params.  Needed when params were loaded from disk AFTER engine
        construction without a matching optimizer-state file — otherwise the
        stale master would overwrite the loaded weights on the first step."""
        r = self.comm.rank
        for bkt in self.buckets:
            src = self.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
            self.master[bkt.shard_off : bkt.shard_off + bkt.per].copy_(src.float())

    def state_dict(self) -> Dict:
        out = {
            "step": self.step_count,
            "world": self.comm.world,
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }
        if self.avg is not None:
            out["avg"] = self.avg
        return out

    def load_state_dict(self, state: Dict) -> None:
        saved_world = int(state.get("world", self.comm.world))
        if saved_world != self.comm.world:
            raise ValueError(
                f"optimizer shard was saved at world_size={saved_world} but "
                f"this run has world_size={self.comm.world} — the ZeRO-1 "
                f"shard layout depends on world size; resume with the same "
                f"--n-workers, or delete the optim.rank*.pt files to resume "
                f"from params only"
            )
        self.step_count = int(state["step"])
        self.master.copy_(state["master"])
        self.exp_avg.copy_(state["exp_avg"])
        self.exp_avg_sq.copy_(state["exp_avg_sq"])
        if self.avg is not None:
            if "avg" in state:
                self.avg.copy_(state["avg"])
            else:
                # old checkpoint without the average: restart it from the
                # restored master rather than the stale init-time clone
                # (which would dominate the running mean at large step_count)
                self.avg.copy_(self.master)
        r = self.comm.rank
        for bkt in self.buckets:
            dst = self.flat_param[bkt.start + r * bkt.per : bkt.start + (r + 1) * bkt.per]
            dst.copy_(self.master[bkt.shard_off : bkt.shard_off + bkt.per].to(self.dtype))
        for bkt in self.buckets:
            self._gather_bucket(bkt)
