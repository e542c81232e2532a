"""Synchronous data-parallel gradient all-reduce over RCCL/xGMI.

Replaces the reference's asynchronous TF parameter-server distribution
(`clusterone_config.py:106-125`, variables sharded onto ps hosts, workers
pushing gradients over gRPC) with the MI355X-idiomatic scheme (SURVEY.md
§5.8): one process per GPU, torch.distributed with the nccl(=RCCL) backend,
and bucketed all-reduce of gradients overlapped with backward.

Bucketing is tuned for xGMI, not NVSwitch: each of the 8 GPUs has 7
point-to-point links (~153 GB/s each), ring all-reduce is per-link bound, so
fewer/larger buckets (default 16 MB) amortize latency; buckets launch as
soon as their last gradient materializes during backward (post-accumulate
hooks), on whatever stream the backend manages, and `finish_backward()`
waits + writes the averaged gradients back.

Overlap with the fused-BPTT flagship path: the decoder backward is split
(sat_amd/models/bptt.py) so the decode-MLP weight grads — the two largest
trainable tensors — accumulate BEFORE the recurrent reverse loop runs.
Buckets are therefore segregated by phase (decode-head params never share
a bucket with recurrent-core params): the head buckets' all-reduces launch
immediately and overlap the rest of backward instead of waiting for the
whole step's gradients.

Works identically over gloo on CPU (the multi-process CI path).
"""

import torch
import torch.distributed as dist


def _phase_key(name):
    """Coarse backward-phase of a param: decode-head grads materialize
    first (DecodeHeadBPTT.backward), everything else at the end of the
    recurrent core's backward."""
    return 'head' if 'dec_fc' in name else 'core'


class _Bucket:
    __slots__ = ('params', 'numel', 'flat', 'offsets', 'pending', 'work',
                 'unfilled', 'completed_in_capture')

    def __init__(self):
        self.params = []
        self.numel = 0
        self.flat = None
        self.offsets = {}
        self.pending = 0
        self.work = None
        self.unfilled = set()
        self.completed_in_capture = False


class DataParallelGrads(object):
    def __init__(self, model, bucket_mb=16, process_group=None):
        self.group = process_group
        self.world = dist.get_world_size(process_group)
        self.buckets = []
        self._param_bucket = {}
        # defer_comm: hooks fill/scale the flat buffers but do NOT launch
        # collectives — launch_deferred_comm() issues them later.  Used
        # by the engine's split-capture mode: the fwd+bwd graph captures
        # the bucket copies while the RCCL calls stay outside capture.
        self.defer_comm = False

        named = [(n, p) for n, p in model.named_parameters()
                 if p.requires_grad]
        # backward produces gradients roughly in reverse parameter order:
        # bucket in reverse so each bucket fills contiguously in time.
        named = list(reversed(named))

        cap = int(bucket_mb * (1 << 20) / 4)  # fp32 elements per bucket
        cur = _Bucket()
        cur_key = None
        for name, p in named:
            key = _phase_key(name)
            if cur.numel and (cur.numel + p.numel() > cap
                              or key != cur_key):
                self.buckets.append(cur)
                cur = _Bucket()
            cur_key = key
            cur.offsets[p] = cur.numel
            cur.params.append(p)
            cur.numel += p.numel()
        if cur.numel:
            self.buckets.append(cur)

        for b in self.buckets:
            for p in b.params:
                self._param_bucket[p] = b
                p.register_post_accumulate_grad_hook(self._hook)
        self.reset()

    def reset(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
            b.unfilled = set(b.params)

    def _hook(self, p):
        b = self._param_bucket[p]
        if b.flat is None or b.flat.device != p.grad.device:
            b.flat = torch.zeros(b.numel, dtype=torch.float32,
                                 device=p.grad.device)
        off = b.offsets[p]
        b.flat[off:off + p.numel()].copy_(
            p.grad.detach().reshape(-1).float())
        b.unfilled.discard(p)
        b.pending -= 1
        if b.pending == 0:
            b.flat.div_(self.world)
            if not self.defer_comm:
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)

    # ---- deferred-comm protocol (engine split-capture mode) ----
    # The fwd+bwd hipGraph captures the bucket fill/scale kernels; the
    # collectives run EAGERLY between graph replay and the optimizer.
    # Capture-static graphs fill the same buckets every replay, so the
    # capture-time bookkeeping (which buckets completed) holds for
    # every replay.

    def snapshot_capture_state(self):
        for b in self.buckets:
            b.completed_in_capture = (b.pending == 0)

    def launch_deferred_comm(self):
        for b in self.buckets:
            if b.flat is None:
                b.flat = torch.zeros(
                    b.numel, dtype=torch.float32,
                    device=next(iter(b.offsets)).device)
            if not b.completed_in_capture:
                # hook-side div_ never ran for this bucket
                b.flat.div_(self.world)
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def wait_deferred_comm(self):
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            for p in b.params:
                if p.grad is None:
                    continue
                off = b.offsets[p]
                g = b.flat[off:off + p.numel()].reshape(p.shape)
                p.grad.detach().copy_(g.to(p.dtype))

    def finish_backward(self):
        """Wait for in-flight buckets and write averaged grads back."""
        for b in self.buckets:
            if b.pending != 0:
                # Gradients never materialized for some params this step
                # (e.g. conditionally-unused layers); reduce the bucket
                # anyway so ranks stay in collective lockstep — but ZERO
                # the unfilled slots first: `flat` still holds the
                # previous step's reduced values for them (ghost grads).
                if b.flat is None:
                    b.flat = torch.zeros(
                        b.numel, dtype=torch.float32,
                        device=next(iter(b.offsets)).device)
                else:
                    for p in b.unfilled:
                        off = b.offsets[p]
                        b.flat[off:off + p.numel()].zero_()
                b.flat.div_(self.world)
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            for p in b.params:
                # A param whose hook never fired gets the cross-rank
                # average written back only if it had a grad to begin
                # with (another rank may have used the layer); params
                # with no grad anywhere stay grad-less so the optimizer
                # sees them as untouched instead of decaying momentum
                # with fabricated zeros.
                if p in b.unfilled and p.grad is None:
                    continue
                off = b.offsets[p]
                g = b.flat[off:off + p.numel()].reshape(p.shape)
                if p.grad is None:
                    p.grad = g.to(p.dtype).clone()
                else:
                    p.grad.detach().copy_(g.to(p.dtype))
        self.reset()
