"""ShardedAdamW: flat-buffer data-parallel training engine.

Replaces the reference's shard_gpt FSDP policy + optax chain
(reference src/model.py:167-178, src/train.py:147-159) with an explicit
MI355X-first design:

- ALL parameters live as views into ONE flat compute-dtype (bf16) buffer
  ``flat_w``; gradients accumulate into views of ONE flat buffer ``flat_g``.
  A collective is therefore a few bucketed RCCL calls on flat ranges, not a
  per-tensor sequence.
- fp32 master weights and Adam state are SHARDED across ranks (ZeRO).
  In ``zero`` mode (shard_model=True) the per-step flow is:
      per microstep: bucketed reduce-scatter(flat_g bf16), each bucket
          launched FROM A BACKWARD HOOK the moment its last grad lands, on
          a side HIP stream -> fp32 shard accumulator
      step: global-norm clip (scalar all-reduce) -> fused AdamW on the shard
            (writes the bf16 shard) -> all-gather(flat_w bf16)
  With 288 GB HBM3E per GPU the full bf16 working weights stay RESIDENT on
  every rank (7B = 14 GB), so there is no per-layer gather on the critical
  path: one all-gather per step, one (bucketed) reduce-scatter per
  microstep — strictly less communication than gather-per-layer ZeRO-3
  under per-block remat.
- ``ddp`` mode (shard_model=False): bucketed all-reduce from the same
  hooks, replicated update.

Bucketing: the padded flat buffer is cut into NB (env MIDGPT_ZERO_BUCKETS,
default 4) ranges whose sizes are multiples of world, so bucket boundaries
need NOT align with parameter boundaries and the flat layout/param offsets
are IDENTICAL to the unbucketed layout. Rank r's shard is the
concatenation of its per-bucket pieces (``self.pieces``); the optimizer
math is elementwise so the concatenated order is irrelevant. Buckets are
launched in strictly DESCENDING index order (backward produces grads in
reverse registration order), which also guarantees an identical collective
launch order on every rank.

Numerics parity with the reference step (src/train.py:79-97):
bf16 grads reduced across ranks, accumulated in fp32 per microbatch;
clip_by_global_norm(1.0) -> Adam(b1=.9, b2=cfg) -> + (wd/lr_peak)*theta
-> *lr_t -> descend.
"""
from __future__ import annotations

import os

import torch

from midgpt_amd import ops
from midgpt_amd.parallel import dist as pdist


class ShardedAdamW:
    def __init__(self, model: torch.nn.Module, *,
                 compute_dtype: torch.dtype = torch.bfloat16,
                 zero: bool = True,
                 beta1: float = 0.9, beta2: float = 0.95,
                 eps: float = 1e-8, weight_decay: float = 1e-4,
                 peak_lr: float = 1e-3, grad_clip: float = 1.0,
                 device: torch.device | None = None,
                 n_buckets: int | None = None):
        self.model = model
        self.beta1, self.beta2, self.eps = beta1, beta2, eps
        self.wd_over_peak = weight_decay / peak_lr
        self.grad_clip = grad_clip
        self.step_count = 0  # number of optimizer steps taken

        self.rank = pdist.get_rank()
        self.world = pdist.get_world_size()
        self.zero = zero and self.world > 1

        params = list(model.parameters())
        if device is None:
            device = params[0].device
        self.device = device
        self.compute_dtype = compute_dtype

        self.meta = []  # (name, shape, offset, numel)
        off = 0
        names = {p: n for n, p in model.named_parameters()}
        for p in params:
            n = p.numel()
            self.meta.append((names[p], tuple(p.shape), off, n))
            off += n
        total = off
        self.total = total
        self.shard_size = (total + self.world - 1) // self.world if self.zero else total
        self.padded = self.shard_size * self.world if self.zero else total
        self.shard_off = self.rank * self.shard_size if self.zero else 0

        # ---- bucket layout: NB flat ranges, sizes % world == 0 ----------
        if n_buckets is None:
            n_buckets = int(os.environ.get("MIDGPT_ZERO_BUCKETS", "4"))
        S = self.shard_size
        nb = max(1, min(n_buckets, S)) if self.zero else \
            max(1, min(n_buckets, max(1, self.padded // (1 << 20))))
        if self.zero:
            cuts = [S * i // nb for i in range(nb + 1)]
            self.buckets = [(cuts[i] * self.world,
                             (cuts[i + 1] - cuts[i]) * self.world)
                            for i in range(nb)]
            # rank's piece of bucket b in FULL-buffer coords, and its offset
            # inside the concatenated shard
            self.pieces = []
            po = 0
            for (o_b, n_b) in self.buckets:
                p_b = n_b // self.world
                self.pieces.append((o_b + self.rank * p_b, p_b, po))
                po += p_b
        else:
            cuts = [self.padded * i // nb for i in range(nb + 1)]
            self.buckets = [(cuts[i], cuts[i + 1] - cuts[i]) for i in range(nb)]
            self.pieces = [(o, n, o) for (o, n) in self.buckets]
        self.n_buckets = len(self.buckets)

        # Full fp32 image (staging) -> master shard + bf16 working buffer.
        full32 = torch.zeros(self.padded, dtype=torch.float32, device=device)
        for p, (_, _, o, n) in zip(params, self.meta):
            full32[o:o + n].copy_(p.detach().reshape(-1).to(device, torch.float32))
        self.master = self._take_shard(full32)
        self.flat_w = full32.to(compute_dtype)
        del full32
        # DOUBLE flat grad buffers on GPU: microstep i's reduction runs on a
        # side HIP stream while microstep i+1's forward/backward fills the
        # other buffer (comm/compute overlap). CPU keeps one buffer with
        # synchronous collectives.
        self._overlap = device.type == "cuda"
        nbuf = 2 if self._overlap else 1
        self.flat_gs = [torch.zeros(self.padded, dtype=compute_dtype,
                                    device=device) for _ in range(nbuf)]
        self._cur = 0
        self.params = params
        # Rebind parameters as views into the flat weight buffer; grads are
        # bound to the current flat grad buffer by _rebind_grads.
        for p, (_, shape, o, n) in zip(params, self.meta):
            p.data = self.flat_w[o:o + n].view(shape)
        self._rebind_grads()

        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)
        # fp32 microbatch-grad accumulator over the shard (zeros each step)
        self.g32 = torch.zeros_like(self.master)
        # bf16 reduce-scatter landing buffer (concatenated piece order)
        self._g16_shards = [torch.empty(self.shard_size, dtype=compute_dtype,
                                        device=device) for _ in range(nbuf)] \
            if self.zero else None
        # bf16 optimizer output shard (contiguous; all-gathered per bucket)
        self._w16_shard = torch.empty(self.shard_size, dtype=compute_dtype,
                                      device=device) if self.zero else None
        if self._overlap:
            self._comm_stream = torch.cuda.Stream(device=device)
            self._pending = []  # [(done_event, buf_index), ...]
        if self.world > 1:
            # force communicator init with a default-stream collective NOW:
            # the first real collective fires from a backward hook on the
            # side stream, which must never be the lazy-rendezvous path
            pdist.all_reduce_(torch.zeros(1, device=device))
        # ---- backward-hook bucket machinery (GPU only) ------------------
        # bucket b covers flat range [o_b, o_b+n_b); a param contributes to
        # every bucket its range overlaps. Buckets launch in descending
        # index order once ready (identical order on all ranks).
        self._bucket_params = [0] * self.n_buckets
        self._param_buckets: list[list[int]] = []
        for (_, _, o, n) in self.meta:
            bl = [b for b, (o_b, n_b) in enumerate(self.buckets)
                  if o < o_b + n_b and o + n > o_b]
            self._param_buckets.append(bl)
            for b in bl:
                self._bucket_params[b] += 1
        self._hooks_on = self._overlap and self.n_buckets > 1
        self._counts = list(self._bucket_params)
        self._ready = [False] * self.n_buckets
        self._next_launch = self.n_buckets - 1
        self._launched = [False] * self.n_buckets
        if self._hooks_on:
            for i, p in enumerate(self.params):
                p.register_post_accumulate_grad_hook(self._make_hook(i))

    # ------------------------------------------------------------------
    def _take_shard(self, full: torch.Tensor) -> torch.Tensor:
        """Concatenated per-bucket pieces of ``full`` for this rank."""
        if not self.zero:
            return full.clone()
        out = torch.empty(self.shard_size, dtype=full.dtype, device=full.device)
        for (fo, p_b, po) in self.pieces:
            out[po:po + p_b].copy_(full[fo:fo + p_b])
        return out

    def _scatter_shard(self, shard: torch.Tensor, full: torch.Tensor):
        """Inverse of _take_shard: place this rank's pieces into ``full``."""
        for (fo, p_b, po) in self.pieces:
            full[fo:fo + p_b].copy_(shard[po:po + p_b])

    @property
    def flat_g(self):
        """The flat grad buffer autograd currently accumulates into."""
        return self.flat_gs[self._cur]

    def _rebind_grads(self):
        cur = self.flat_gs[self._cur]
        for p, (_, shape, o, n) in zip(self.params, self.meta):
            p.grad = cur[o:o + n].view(shape)

    # ------------------------------------------------------------------
    # bucket launch (from hooks during backward, or microstep_end catch-up)
    # ------------------------------------------------------------------
    def _make_hook(self, param_idx: int):
        def hook(_p):
            for b in self._param_buckets[param_idx]:
                self._counts[b] -= 1
                if self._counts[b] == 0:
                    self._ready[b] = True
            while self._next_launch >= 0 and self._ready[self._next_launch]:
                self._launch_bucket(self._next_launch)
                self._next_launch -= 1
        return hook

    def _launch_bucket(self, b: int):
        """Enqueue bucket b's collective on the comm stream (GPU path)."""
        cur = self._cur
        o_b, n_b = self.buckets[b]
        if n_b == 0:
            self._launched[b] = True
            return
        ready = torch.cuda.Event()
        ready.record()  # this bucket's grads complete on the compute stream
        with torch.cuda.stream(self._comm_stream):
            self._comm_stream.wait_event(ready)
            if self.zero:
                fo, p_b, po = self.pieces[b]
                pdist.reduce_scatter_flat(
                    self.flat_gs[cur][o_b:o_b + n_b],
                    self._g16_shards[cur][po:po + p_b])
            elif self.world > 1:
                pdist.all_reduce_(self.flat_gs[cur][o_b:o_b + n_b])
        self._launched[b] = True

    def _drain(self, buf: int | None = None):
        """Wait for pending side-stream reductions (all, or just ``buf``'s)
        and fold them into the fp32 accumulator on the compute stream."""
        keep = []
        for done, b in self._pending:
            if buf is not None and b != buf:
                keep.append((done, b))
                continue
            torch.cuda.current_stream().wait_event(done)
            if self.zero:
                self.g32.add_(self._g16_shards[b])
            else:
                self.g32.add_(self.flat_gs[b])
            self.flat_gs[b].zero_()
        self._pending = keep

    # ------------------------------------------------------------------
    # per-microstep: called after each microbatch backward
    # ------------------------------------------------------------------
    def microstep_end(self):
        """Finish this microbatch's grad reduction and swap buffers
        (reference microstep parity: src/train.py:85-92).

        GPU: each bucket's collective was already ENQUEUED on the side HIP
        stream from its backward hook (overlapping the rest of backward);
        here any stragglers launch, a completion event is recorded, and the
        grad views swap to the other flat buffer so the tail of the
        reduction overlaps the next microbatch's forward/backward. The
        pipeline — events, buffer swap, view rebinding, deferred
        accumulation — runs identically at world=1, so single-GPU tests and
        the bench exercise the full path; only the collective call itself
        needs more ranks."""
        if not self._overlap:  # CPU: synchronous, still bucketed
            for b, (o_b, n_b) in enumerate(self.buckets):
                if n_b == 0:
                    continue
                if self.zero:
                    fo, p_b, po = self.pieces[b]
                    pdist.reduce_scatter_flat(
                        self.flat_gs[0][o_b:o_b + n_b],
                        self._g16_shards[0][po:po + p_b])
                elif self.world > 1:
                    pdist.all_reduce_(self.flat_gs[0][o_b:o_b + n_b])
            if self.zero:
                self.g32.add_(self._g16_shards[0])
            else:
                self.g32.add_(self.flat_gs[0])
            self.flat_gs[0].zero_()
            return
        cur = self._cur
        if self._hooks_on:
            # launch any buckets whose hooks did not fire (e.g. a param
            # without grad this microstep), in descending order
            while self._next_launch >= 0:
                self._launch_bucket(self._next_launch)
                self._next_launch -= 1
            # reset hook state for the next microstep
            self._counts = list(self._bucket_params)
            self._ready = [False] * self.n_buckets
            self._launched = [False] * self.n_buckets
            self._next_launch = self.n_buckets - 1
        else:
            for b in range(self.n_buckets - 1, -1, -1):
                self._launch_bucket(b)
            self._launched = [False] * self.n_buckets
        done = torch.cuda.Event()
        with torch.cuda.stream(self._comm_stream):
            done.record(self._comm_stream)
        self._pending.append((done, cur))
        # swap to the other buffer for the next microbatch
        self._cur = cur ^ 1
        self._drain(buf=self._cur)  # its previous reduction must be consumed
        self._rebind_grads()

    # ------------------------------------------------------------------
    # per-step
    # ------------------------------------------------------------------
    def defer_until(self, evt) -> None:
        """Make the next optimizer step wait on ``evt`` before mutating
        master/m/v (used by the async checkpoint D2H snapshot)."""
        self._defer_evt = evt

    def step(self, lr: float, g_accum_iters: int = 1) -> torch.Tensor:
        """Apply one optimizer step. Returns a device scalar tensor from
        which the pre-clip global grad norm is ``sqrt(t) * scale`` (no host
        sync on the step path). ``lr`` is this step's scheduled LR."""
        if getattr(self, "_defer_evt", None) is not None:
            torch.cuda.current_stream().wait_event(self._defer_evt)
            self._defer_evt = None
        if self._overlap:
            self._drain()
        scale = 1.0 / (g_accum_iters * self.world)
        sq = self.g32.pow(2).sum()
        if self.zero:
            pdist.all_reduce_(sq)
        self.step_count += 1
        out_view = (self._w16_shard if self.zero else self.flat_w)
        ops.adamw_step(self.master, self.g32, self.m, self.v,
                       out_view if self.compute_dtype == torch.bfloat16 else None,
                       lr=lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                       wd_over_peak_lr=self.wd_over_peak,
                       grad_scale=scale, clip_norm=self.grad_clip,
                       sq_sum=sq, step=self.step_count)
        if self.compute_dtype != torch.bfloat16:
            out_view.copy_(self.master.to(self.compute_dtype))
        if self.zero:
            self._allgather_weights()
        self.g32.zero_()
        return sq

    def _allgather_weights(self):
        for b, (o_b, n_b) in enumerate(self.buckets):
            if n_b == 0:
                continue
            fo, p_b, po = self.pieces[b]
            pdist.all_gather_flat(self.flat_w[o_b:o_b + n_b],
                                  self._w16_shard[po:po + p_b])

    # ------------------------------------------------------------------
    # checkpoint interface
    # ------------------------------------------------------------------
    def state_shard(self) -> dict:
        """This rank's checkpoint shard (fp32 master + Adam state)."""
        return {
            "master": self.master,
            "m": self.m,
            "v": self.v,
            "step_count": self.step_count,
            "shard_off": self.shard_off,
            "shard_size": self.shard_size,
            # piece map: shard[po:po+n] lives at full[fo:fo+n]
            "pieces": [[fo, p_b, po] for (fo, p_b, po) in self.pieces],
            "total": self.total,
            "padded": self.padded,
            "world": self.world if self.zero else 1,
        }

    def load_state_full(self, full_master, full_m, full_v, step_count: int):
        """Load from FULL (padded or unpadded) fp32 state tensors; reshards
        for the current world size and bucket layout."""
        def take(t):
            out = torch.zeros(self.padded, dtype=torch.float32)
            k = min(t.numel(), self.padded)
            out[:k] = t.reshape(-1)[:k].float()
            return self._take_shard(out.to(self.device))
        self.master.copy_(take(full_master))
        self.m.copy_(take(full_m))
        self.v.copy_(take(full_v))
        self.step_count = step_count
        self.refresh_weights()

    def refresh_weights(self):
        """Recompute the bf16 working weights from master (after restore)."""
        if self.zero:
            self._w16_shard.copy_(self.master.to(self.compute_dtype))
            self._allgather_weights()
        else:
            self.flat_w.copy_(self.master.to(self.compute_dtype))

    def named_param_manifest(self) -> list[dict]:
        """Ordered named-parameter manifest (the flat-leaves contract of the
        reference checkpoints, src/train.py:215, keyed by name+offset)."""
        return [{"name": n, "shape": list(s), "offset": o, "numel": k}
                for (n, s, o, k) in self.meta]
