"""Fused data-parallel optimizers — the MI355X flagship gradient path.

This is the structural redesign of the reference's hot loop (SURVEY.md
§3.2/§3.3).  The reference per step: extract grads → batched pack kernel →
cast fp16 → ncclAllReduce → fused cast-back+scale → unpack → per-param
optimizer kernels (≈161 launches).  Here, by construction:

* every parameter lives in one flat low-precision arena (bf16); grads
  either bind as views into a flat bf16 wire arena (bucketed mode — the
  per-bucket all-reduce runs in place during backward) or stay with
  autograd and are gathered by ONE batched pack kernel at step()
  (``grad_pack``, the GPU default: binding views makes AccumulateGrad
  launch a per-param add kernel every step — 177 launches / 3.4 ms on
  ResNet-50 b512 — while the K1 gather is one 51 MB-read kernel);
* the all-reduce is RCCL over xGMI, in place on the grad arena, on a side
  HIP stream, double-buffered (grads of step n reduce while step n+1
  computes — the reference's _DoubleBufferingOptimizer semantics, 1-step
  stale, without its worker thread);
* the whole-model update is ONE fused CDNA4 kernel (optim.hip): fp32
  master params + momentum + mean-scale + bf16 mirror write in a single
  HBM pass.

Per step: 1 RCCL all-reduce (overlapped) + 1 fused kernel.  The wire
volume for ResNet-50 is 51 MB bf16 (reference: 102 MB fp32 or 51 MB fp16
plus two extra full pack/unpack HBM round-trips).
"""

import torch
import torch.distributed as dist

from chainermn_amd import ops


def _is_cl(t):
    return (t.dim() == 4 and not t.is_contiguous()
            and t.is_contiguous(memory_format=torch.channels_last))


def _arena_view(flat, like):
    """View a flat arena slice with ``like``'s shape AND memory format.

    channels_last params keep their strides (conv kernel selection on
    MI355X depends on it); the arena then holds the param's PHYSICAL
    (NHWC) element order, which is also what ``_physical_flat`` yields.
    """
    if _is_cl(like):
        n, c, h, w = like.shape
        return flat.as_strided(like.shape, (h * w * c, 1, w * c, c))
    return flat.view_as(like)


def _physical_flat(t):
    """``t``'s elements in memory order — the arena layout of
    ``_arena_view`` — regardless of memory format."""
    if _is_cl(t):
        return t.permute(0, 2, 3, 1).reshape(-1)
    return t.reshape(-1)


class _FlatState(object):
    """Flattens a model's params into arenas and rebinds p.data/p.grad."""

    def __init__(self, model, comm, grad_dtype=torch.bfloat16,
                 n_grad_buffers=2):
        params = [p for _, p in sorted(model.named_parameters())
                  if p.requires_grad]
        if not params:
            raise ValueError('model has no trainable parameters')
        dev = params[0].device
        dtype = params[0].dtype
        for p in params:
            if p.dtype != dtype:
                raise ValueError('FusedDataParallel requires uniform '
                                 'parameter dtype')
        self.comm = comm
        self.device = dev
        self.model_dtype = dtype
        # autograd accumulates into p.grad views, so the grad (wire) arena
        # must match the param dtype — bf16 params give bf16 wire traffic
        # for free; fp32 models reduce fp32 (compressed-wire fp32 models go
        # through PureRcclCommunicator's pack+cast path instead).
        if grad_dtype != dtype:
            grad_dtype = dtype
        self.grad_dtype = grad_dtype
        self.plan = ops.PackPlan(params, dev)
        n = self.plan.total
        # fp32 master + low-precision mirror (or fp32-only when the model
        # itself is fp32)
        self.master = torch.zeros(n, dtype=torch.float32, device=dev)
        self.lowp = None
        if dtype != torch.float32:
            self.lowp = torch.zeros(n, dtype=dtype, device=dev)
        self.grads = [torch.zeros(n, dtype=grad_dtype, device=dev)
                      for _ in range(n_grad_buffers)]
        self.params = params
        # move params into the arena
        with torch.no_grad():
            for p, off, sz in zip(params, self.plan.offsets_cpu,
                                  self.plan.sizes_cpu):
                self.master[off:off + sz].copy_(
                    _physical_flat(p.data).float())
                arena = self.lowp if self.lowp is not None else self.master
                view = _arena_view(arena[off:off + sz], p.data)
                view.copy_(p.data)
                p.data = view
        self.cur = 0
        self._bind_grads(0)

    def _bind_grads(self, idx):
        for p, off, sz in zip(self.params, self.plan.offsets_cpu,
                              self.plan.sizes_cpu):
            p.grad = _arena_view(self.grads[idx][off:off + sz], p.data)

    def unbind_grads(self):
        """Pack-mode: leave p.grad to autograd (no accumulate-into-view
        adds during backward; step() gathers with ONE batched kernel)."""
        for p in self.params:
            p.grad = None

    def pack_into(self, idx):
        """Gather every p.grad into arena ``idx`` (one kernel), then drop
        the autograd-owned grads.  Grad memory order must match the
        arena's physical order (channels_last conv grads come back
        channels_last; normalize if some op produced a different
        layout)."""
        for p in self.params:
            g = p.grad
            if g is not None and g.stride() != p.data.stride():
                p.grad = g.contiguous() if p.data.is_contiguous() \
                    else g.contiguous(memory_format=torch.channels_last)
        ops.pack_grads(self.plan, self.grads[idx])
        for p in self.params:
            p.grad = None

    def grad_arena(self, idx=None):
        return self.grads[self.cur if idx is None else idx]

    def sync_master_from_params(self):
        """Refresh fp32 master after an external param mutation
        (e.g. bcast_data)."""
        with torch.no_grad():
            for p, off, sz in zip(self.params, self.plan.offsets_cpu,
                                  self.plan.sizes_cpu):
                self.master[off:off + sz].copy_(
                    _physical_flat(p.data).float())


class _BucketEngine(object):
    """Zero-staleness comm/compute overlap: the flat grad arena is split
    into contiguous buckets; per-param post-accumulate-grad hooks launch
    each bucket's in-place all-reduce on the comm stream as soon as the
    bucket's last gradient lands during backward (the modern alternative
    to the reference's 1-step-stale double buffering).  Buckets are
    launched in a FIXED order (a ready-prefix walk over reverse arena
    order, which tracks backward's param order) so every rank issues RCCL
    collectives identically — no cross-rank deadlock by construction.
    """

    def __init__(self, state, comm, bucket_mb=25):
        self.state = state
        self.comm = comm
        arena = state.grad_arena()
        esize = arena.element_size()
        bucket_elems = max(int(bucket_mb * (1 << 20)) // esize, 1)
        # buckets over slot ranges; params assigned by start offset
        spans = list(zip(state.plan.offsets_cpu, state.plan.sizes_cpu))
        buckets = []
        cur_params, cur_start = [], None
        for i, (off, sz) in enumerate(spans):
            if cur_start is None:
                cur_start = off
            cur_params.append(i)
            end = (off + sz + ops.ALIGN - 1) // ops.ALIGN * ops.ALIGN
            if end - cur_start >= bucket_elems or i == len(spans) - 1:
                buckets.append((cur_start, end, list(cur_params)))
                cur_params, cur_start = [], None
        # reverse arena order approximates backward completion order
        self.buckets = list(reversed(buckets))
        self.param_bucket = {}
        for bi, (_, _, pidx) in enumerate(self.buckets):
            for i in pidx:
                self.param_bucket[id(state.params[i])] = bi
        self.use_gpu = arena.is_cuda
        if self.use_gpu:
            self.stream = torch.cuda.Stream()
            self.events = [torch.cuda.Event() for _ in self.buckets]
        self._reset()
        # NOTE: hooks live as long as the params; building a second
        # optimizer over the same model is unsupported (as is re-arenaing
        # params via a second _FlatState).
        self._hook_handles = [p.register_post_accumulate_grad_hook(
            self._hook) for p in state.params]

    def _reset(self):
        self.pending = [len(b[2]) for b in self.buckets]
        self.launched = 0  # buckets [0, launched) already issued

    # --- overlap telemetry -------------------------------------------
    # If backward completion order diverges from reverse arena order
    # (branchy models), ready buckets pile up behind a not-yet-ready
    # prefix bucket and flush synchronously at finish() — correct but
    # with no overlap.  Count it and warn once so the degradation is
    # visible instead of silent.
    flushed_at_finish = 0
    _warned_flush = False

    def _hook(self, p):
        bi = self.param_bucket[id(p)]
        if bi < self.launched or self.pending[bi] <= 0:
            raise RuntimeError(
                'bucketed overlap saw a second gradient for a bucket that '
                'already reduced: multi-backward gradient accumulation is '
                'not supported with overlap="bucketed" — use '
                'overlap=False or overlap=True (double buffering)')
        self.pending[bi] -= 1
        self._launch_ready_prefix()

    def _launch_ready_prefix(self):
        while self.launched < len(self.buckets) \
                and self.pending[self.launched] == 0:
            start, end, _ = self.buckets[self.launched]
            shard = self.state.grad_arena()[start:end]
            if self.comm.size > 1:
                if self.use_gpu:
                    main = torch.cuda.current_stream()
                    self.stream.wait_stream(main)
                    with torch.cuda.stream(self.stream):
                        dist.all_reduce(shard, group=self.comm.group)
                        self.events[self.launched].record(self.stream)
                else:
                    dist.all_reduce(shard, group=self.comm.group)
            elif self.use_gpu:
                self.events[self.launched].record(
                    torch.cuda.current_stream())
            self.launched += 1

    def finish(self):
        """Flush un-launched buckets, wait for all; call before update."""
        n_flush = len(self.buckets) - self.launched
        if n_flush > 0:
            self.flushed_at_finish += n_flush
            if not self._warned_flush and len(self.buckets) > 1:
                self._warned_flush = True
                import warnings
                warnings.warn(
                    'bucketed overlap: %d of %d buckets flushed '
                    'synchronously at finish() (backward completion '
                    'order diverges from arena order, or some params '
                    'got no gradient) — training is correct but those '
                    'buckets did not overlap with backward; see '
                    '_BucketEngine.flushed_at_finish' %
                    (n_flush, len(self.buckets)))
        for bi in range(self.launched, len(self.buckets)):
            self.pending[bi] = 0
        self._launch_ready_prefix()
        if self.use_gpu:
            main = torch.cuda.current_stream()
            for ev in self.events:
                main.wait_event(ev)
        self._reset()


class FusedDataParallelSGD(object):
    """SGD/MomentumSGD over flat arenas with overlapped bf16 all-reduce.

    ``overlap=True`` → double-buffered (1-step-stale) gradients: the
    all-reduce of step n runs on a side stream under the fwd/bwd of step
    n+1; the first step applies no update (the reference's flagship
    double-buffering config).  ``overlap='bucketed'`` → zero-staleness:
    per-bucket all-reduce overlapped WITHIN backward via grad hooks.
    ``overlap=False`` → synchronous.
    """

    def __init__(self, model, comm, lr, momentum=0.9, weight_decay=0.0,
                 dampening=0.0, nesterov=False, overlap=True,
                 grad_dtype=torch.bfloat16, bucket_mb=25, grad_pack=None):
        self.comm = comm
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.dampening = dampening
        self.nesterov = nesterov
        self.overlap = overlap
        double_buf = overlap is True
        self.state = _FlatState(model, comm,
                                grad_dtype=grad_dtype,
                                n_grad_buffers=2 if double_buf else 1)
        n = self.state.plan.total
        self.v = torch.zeros(n, dtype=torch.float32,
                             device=self.state.device) \
            if momentum else None
        self.buckets = None
        if overlap == 'bucketed':
            self.buckets = _BucketEngine(self.state, comm,
                                         bucket_mb=bucket_mb)
        elif double_buf:
            self.stream = torch.cuda.Stream()
            self.ev_ar = [torch.cuda.Event(), torch.cuda.Event()]
        # grad_pack: leave p.grad to autograd during backward and gather
        # with ONE batched kernel at step() instead of binding arena
        # views (which makes AccumulateGrad launch one add kernel per
        # param per step).  Default on for GPU non-bucketed modes
        # (bucketed needs the views for in-place per-bucket all-reduce
        # during backward).  Pass grad_pack=False under hipGraph capture:
        # the per-step pointer upload is a pageable H2D copy, which
        # capture rejects.
        if grad_pack is None:
            grad_pack = (overlap != 'bucketed'
                         and self.state.device.type == 'cuda')
        self.grad_pack = grad_pack
        if grad_pack:
            self.state.unbind_grads()
        self._step = 0

    def zero_grad(self, set_to_none=False):
        """No-op after step 0 BY DESIGN: step() zeroes the active arena
        after applying it, and p.grad must stay bound to the arena views
        (``set_to_none=True`` would break the zero-copy wire path, so it
        raises).  If you mutate ``p.grad`` manually between steps, call
        ``zero_grad_force()`` — plain zero_grad will NOT clear the
        arena for you."""
        if self.grad_pack:
            return  # autograd owns the grads; step() packs and clears
        if set_to_none:
            raise ValueError(
                'FusedDataParallel binds p.grad to the flat wire arena; '
                'set_to_none=True is unsupported (pass False)')
        if self._step == 0:
            self.state.grad_arena().zero_()

    def zero_grad_force(self):
        """Explicitly zero the ACTIVE grad arena (after manual p.grad
        mutation; normal training never needs this)."""
        self.state.grad_arena().zero_()

    def _reduce_async(self, idx):
        g = self.state.grads[idx]
        main = torch.cuda.current_stream()
        self.stream.wait_stream(main)
        with torch.cuda.stream(self.stream):
            dist.all_reduce(g, group=self.comm.group)
            self.ev_ar[idx].record(self.stream)

    def _apply(self, g):
        ops.fused_sgd(self.state.master, g, self.v, self.lr,
                      momentum=self.momentum,
                      weight_decay=self.weight_decay,
                      dampening=self.dampening, nesterov=self.nesterov,
                      grad_scale=1.0 / max(self.comm.size, 1),
                      p_lowp=self.state.lowp)

    def step(self):
        st = self.state
        if self.overlap == 'bucketed':
            self.buckets.finish()
            g = st.grad_arena()
            self._apply(g)
            g.zero_()
            self._step += 1
            return
        if not self.overlap:
            g = st.grad_arena()
            if self.grad_pack:
                st.pack_into(st.cur)
            if self.comm.size > 1:
                dist.all_reduce(g, group=self.comm.group)
            self._apply(g)
            if not self.grad_pack:
                g.zero_()
            self._step += 1
            return
        cur, prev = st.cur, 1 - st.cur
        main = torch.cuda.current_stream()
        if self.grad_pack:
            st.pack_into(cur)
        if self.comm.size > 1:
            self._reduce_async(cur)
        else:
            self.ev_ar[cur].record(main)
        if self._step > 0:
            main.wait_event(self.ev_ar[prev])
            self._apply(st.grads[prev])
            if not self.grad_pack:
                st.grads[prev].zero_()
        # next backward accumulates into `prev` arena (views mode)
        if not self.grad_pack:
            st._bind_grads(prev)
        st.cur = prev
        self._step += 1

    def wait(self):
        if self.overlap is True and self._step > 0:
            torch.cuda.current_stream().wait_event(
                self.ev_ar[1 - self.state.cur])

    # torch.optim-ish compat
    @property
    def param_groups(self):
        return [{'lr': self.lr, 'momentum': self.momentum,
                 'weight_decay': self.weight_decay,
                 'params': self.state.params}]

    def state_dict(self):
        return {'step': self._step, 'master': self.state.master,
                'v': self.v, 'lr': self.lr}

    def load_state_dict(self, sd):
        self._step = sd['step']
        self.state.master.copy_(sd['master'])
        if self.v is not None and sd.get('v') is not None:
            self.v.copy_(sd['v'])
        # params mirror the (possibly loaded) master
        if self.state.lowp is not None:
            ops.cast_flat(self.state.master, self.state.lowp)
        else:
            pass  # params ARE master views


class FusedDataParallelAdam(FusedDataParallelSGD):
    """Adam/AdamW over the flat arenas (one fused kernel per step)."""

    def __init__(self, model, comm, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adamw=False, overlap=True,
                 grad_dtype=torch.bfloat16, bucket_mb=25, grad_pack=None):
        super().__init__(model, comm, lr, momentum=0.0,
                         weight_decay=weight_decay, overlap=overlap,
                         grad_dtype=grad_dtype, bucket_mb=bucket_mb,
                         grad_pack=grad_pack)
        n = self.state.plan.total
        dev = self.state.device
        self.betas = betas
        self.eps = eps
        self.adamw = adamw
        self.m = torch.zeros(n, dtype=torch.float32, device=dev)
        self.vv = torch.zeros(n, dtype=torch.float32, device=dev)
        self._adam_step = 0

    def _apply(self, g):
        self._adam_step += 1
        ops.fused_adam(self.state.master, g, self.m, self.vv, self.lr,
                       self.betas[0], self.betas[1], self.eps,
                       weight_decay=self.weight_decay,
                       step=self._adam_step, adamw=self.adamw,
                       grad_scale=1.0 / max(self.comm.size, 1),
                       p_lowp=self.state.lowp)

    def state_dict(self):
        sd = super().state_dict()
        sd.update({'m': self.m, 'vv': self.vv,
                   'adam_step': self._adam_step})
        return sd

    def load_state_dict(self, sd):
        super().load_state_dict(sd)
        self.m.copy_(sd['m'])
        self.vv.copy_(sd['vv'])
        self._adam_step = sd['adam_step']


class FusedDataParallelLARS(FusedDataParallelSGD):
    """Layer-wise adaptive rate scaling (You et al. 2017) over the same
    arenas — the large-batch config (BASELINE.json config 5).  Per layer:
    trust = eta·||p||/(||g|| + wd·||p|| + eps); v = m·v + trust·lr·(g+wd·p).
    """

    def __init__(self, model, comm, lr, momentum=0.9, weight_decay=5e-5,
                 eta=0.001, eps=1e-9, overlap=True,
                 grad_dtype=torch.bfloat16, bucket_mb=25, grad_pack=None):
        super().__init__(model, comm, lr, momentum=momentum,
                         weight_decay=weight_decay, overlap=overlap,
                         grad_dtype=grad_dtype, bucket_mb=bucket_mb,
                         grad_pack=grad_pack)
        if self.v is None:
            n = self.state.plan.total
            self.v = torch.zeros(n, dtype=torch.float32,
                                 device=self.state.device)
        self.eta = eta
        self.eps = eps
        nseg = len(self.state.plan.sizes_cpu)
        self.p_sqn = torch.zeros(nseg, dtype=torch.float32,
                                 device=self.state.device)
        self.g_sqn = torch.zeros(nseg, dtype=torch.float32,
                                 device=self.state.device)

    def _apply(self, g):
        st = self.state
        plan = st.plan
        ops.segment_sq_norms(st.master, plan.offsets, plan.sizes, self.p_sqn)
        ops.segment_sq_norms(g, plan.offsets, plan.sizes, self.g_sqn)
        ops.lars_update(st.master, g, self.v, plan.offsets, plan.sizes,
                        self.p_sqn, self.g_sqn, self.lr, self.momentum,
                        self.weight_decay, self.eta, self.eps,
                        grad_scale=1.0 / max(self.comm.size, 1),
                        p_lowp=st.lowp)
