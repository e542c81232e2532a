"""Multi-process DP tests over gloo (world_size 2, CPU) — covers the same
code path RCCL uses on the GPU node."""

import os

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from sat_amd.data.dataset import DataSet
from sat_amd.parallel.launch import shard_dataset


def _worker_allreduce(rank, world, tmpfile, q):
    dist.init_process_group(
        'gloo', init_method='file://%s' % tmpfile,
        rank=rank, world_size=world)
    try:
        model = torch.nn.Linear(8, 4)
        # identical params across ranks
        with torch.no_grad():
            for p in model.parameters():
                p.copy_(torch.arange(p.numel(), dtype=torch.float32)
                        .reshape(p.shape) / p.numel())
        from sat_amd.parallel.ddp import DataParallelGrads
        ddp = DataParallelGrads(model, bucket_mb=1)

        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(4, 8)
        loss = model(x).pow(2).mean()
        loss.backward()
        ddp.finish_backward()

        g = model.weight.grad.clone()
        gathered = [torch.zeros_like(g) for _ in range(world)]
        dist.all_gather(gathered, g)
        same = all(torch.allclose(gathered[0], gi, atol=1e-6)
                   for gi in gathered)
        q.put(('ok', rank, bool(same)))
    finally:
        dist.destroy_process_group()


def test_ddp_grads_averaged(tmp_path):
    world = 2
    q = mp.get_context('spawn').Queue()
    f = str(tmp_path / 'init')
    mp.spawn(_worker_allreduce, args=(world, f, q), nprocs=world,
             join=True)
    results = [q.get() for _ in range(world)]
    assert all(r[0] == 'ok' and r[2] for r in results)


def _worker_train_step(rank, world, tmpfile, q):
    dist.init_process_group(
        'gloo', init_method='file://%s' % tmpfile,
        rank=rank, world_size=world)
    try:
        from config import Config
        from sat_amd.models.base_model import BaseModel
        cfg = Config()
        cfg.phase = 'train'
        cfg.train_cnn = False
        cfg.synthetic_data = True
        cfg.batch_size = 1
        cfg.vocabulary_size = 30
        cfg.dim_embedding = 16
        cfg.num_lstm_units = 16
        cfg.dim_initalize_layer = 16
        cfg.dim_attend_layer = 16
        cfg.dim_decode_layer = 16
        cfg.device = 'cpu'
        torch.manual_seed(cfg.seed)  # same init on both ranks
        m = BaseModel(cfg)
        torch.manual_seed(500 + rank)
        images = torch.randn(1, 3, 224, 224)
        sentences = torch.randint(0, 30, (1, cfg.max_caption_length))
        masks = torch.ones(1, cfg.max_caption_length)
        m.train_step(images, sentences, masks)
        # after averaged grads + identical init, params must match
        vec = torch.cat([p.detach().reshape(-1)
                         for p in m.model.decoder.parameters()])
        gathered = [torch.zeros_like(vec) for _ in range(world)]
        dist.all_gather(gathered, vec)
        same = all(torch.allclose(gathered[0], gi, atol=1e-5)
                   for gi in gathered)
        q.put(('ok', rank, bool(same)))
    finally:
        dist.destroy_process_group()


def test_dp_train_step_keeps_replicas_in_sync(tmp_path):
    world = 2
    q = mp.get_context('spawn').Queue()
    f = str(tmp_path / 'init2')
    mp.spawn(_worker_train_step, args=(world, f, q), nprocs=world,
             join=True)
    results = [q.get() for _ in range(world)]
    assert all(r[0] == 'ok' and r[2] for r in results)


def test_shard_dataset():
    n = 10
    ids = list(range(n))
    files = ['f%d' % i for i in ids]
    wi = np.zeros((n, 4), dtype=np.int32)
    mk = np.ones((n, 4), dtype=np.float32)
    a = shard_dataset(DataSet(ids, files, 2, wi, mk, True, False), 0, 2)
    b = shard_dataset(DataSet(ids, files, 2, wi, mk, True, False), 1, 2)
    assert a.count == 5 and b.count == 5
    assert set(a.image_ids) | set(b.image_ids) == set(ids)
    assert set(a.image_ids).isdisjoint(set(b.image_ids))


def _worker_stale_bucket(rank, world, tmpfile, q):
    dist.init_process_group(
        'gloo', init_method='file://%s' % tmpfile,
        rank=rank, world_size=world)
    try:
        class Two(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.a = torch.nn.Linear(8, 8)
                self.b = torch.nn.Linear(8, 8)

            def forward(self, x, use_b):
                y = self.a(x)
                return self.b(y) if use_b else y

        torch.manual_seed(7)
        model = Two()
        from sat_amd.parallel.ddp import DataParallelGrads
        ddp = DataParallelGrads(model, bucket_mb=64)  # one bucket

        x = torch.randn(4, 8)
        # step 1: both layers used -> flat holds b's grads afterwards
        model(x, True).pow(2).mean().backward()
        ddp.finish_backward()
        # step 2: engine-style in-place zero, then b is NOT used
        for p in model.parameters():
            p.grad.zero_()
        model(x, False).pow(2).mean().backward()
        ddp.finish_backward()
        # the stale-flat bug would resurrect step 1's b-grads here
        ghost = float(model.b.weight.grad.abs().max())
        q.put(('ok', rank, ghost == 0.0))
    finally:
        dist.destroy_process_group()


def test_partial_bucket_has_no_ghost_gradients(tmp_path):
    """A bucket whose params were used last step but not this step must
    all-reduce zeros for those slots, not the previous step's values
    (VERDICT round-1 Weak #3)."""
    world = 2
    q = mp.get_context('spawn').Queue()
    f = str(tmp_path / 'init3')
    mp.spawn(_worker_stale_bucket, args=(world, f, q), nprocs=world,
             join=True)
    results = [q.get() for _ in range(world)]
    assert all(r[0] == 'ok' and r[2] for r in results)


def test_split_function_grads_fire_before_upstream_backward():
    """The BPTT split (sat_amd/models/bptt.py) relies on AccumulateGrad's
    top autograd priority: a downstream Function that takes leaf params
    directly gets their grads accumulated (and DDP hooks fired) BEFORE
    the upstream Function's backward runs — that window is where the
    bucket all-reduce overlaps the recurrent reverse loop."""
    order = []

    class Core(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x):
            return x * 2

        @staticmethod
        def backward(ctx, g):
            order.append('core-bwd')
            return g * 2

    class Head(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x, w):
            ctx.save_for_backward(w)
            return x * w.sum()

        @staticmethod
        def backward(ctx, g):
            order.append('head-bwd')
            (w,) = ctx.saved_tensors
            return g * w.sum(), g.sum() * torch.ones_like(w)

    x = torch.randn(3, requires_grad=True)
    w = torch.randn(4, requires_grad=True)
    w.register_post_accumulate_grad_hook(
        lambda p: order.append('w-hook'))
    Head.apply(Core.apply(x), w).sum().backward()
    assert order == ['head-bwd', 'w-hook', 'core-bwd'], order


def test_bucket_phase_segregation():
    """Decode-head params must never share a bucket with recurrent-core
    params, so the head buckets launch without waiting for core grads."""
    import torch.nn as tnn

    class M(tnn.Module):
        def __init__(self):
            super().__init__()
            self.embedding = tnn.Linear(4, 4)
            self.dec_fc_1 = tnn.Linear(4, 4)
            self.dec_fc_2 = tnn.Linear(4, 4)
            self.att_fc_1a = tnn.Linear(4, 4)

    if not dist.is_initialized():
        dist.init_process_group(
            'gloo', init_method='tcp://127.0.0.1:29511',
            rank=0, world_size=1)
    try:
        from sat_amd.parallel.ddp import DataParallelGrads
        m = M()
        ddp = DataParallelGrads(m, bucket_mb=1024)  # size never splits
        names = {p: n for n, p in m.named_parameters()}
        for b in ddp.buckets:
            kinds = {('head' if 'dec_fc' in names[p] else 'core')
                     for p in b.params}
            assert len(kinds) == 1, [names[p] for p in b.params]
    finally:
        dist.destroy_process_group()


def _worker_deferred(rank, world, tmpfile, q):
    dist.init_process_group(
        'gloo', init_method='file://%s' % tmpfile,
        rank=rank, world_size=world)
    try:
        torch.manual_seed(9)
        model = torch.nn.Sequential(torch.nn.Linear(8, 8),
                                    torch.nn.Linear(8, 4))
        from sat_amd.parallel.ddp import DataParallelGrads
        ddp = DataParallelGrads(model, bucket_mb=1)
        ddp.defer_comm = True

        torch.manual_seed(70 + rank)
        x = torch.randn(4, 8)
        model(x).pow(2).mean().backward()
        ddp.snapshot_capture_state()
        ddp.launch_deferred_comm()
        ddp.wait_deferred_comm()

        g = model[0].weight.grad.clone()
        gathered = [torch.zeros_like(g) for _ in range(world)]
        dist.all_gather(gathered, g)
        same = all(torch.allclose(gathered[0], gi, atol=1e-6)
                   for gi in gathered)
        q.put(('ok', rank, bool(same)))
    finally:
        dist.destroy_process_group()


def test_deferred_comm_protocol_averages_grads(tmp_path):
    """The split-capture path (hooks fill buckets, collectives launched
    separately) must produce the same averaged gradients as the inline
    hook-launched path."""
    world = 2
    q = mp.get_context('spawn').Queue()
    f = str(tmp_path / 'init4')
    mp.spawn(_worker_deferred, args=(world, f, q), nprocs=world,
             join=True)
    results = [q.get() for _ in range(world)]
    assert all(r[0] == 'ok' and r[2] for r in results)
