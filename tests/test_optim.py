import torch

from config import Config
from sat_amd.optim import Optimizer


def _cfg(kind='Adam', clip=1e9):
    cfg = Config()
    cfg.optimizer = kind
    cfg.clip_gradients = clip
    cfg.initial_learning_rate = 0.01
    return cfg


def test_adam_matches_torch():
    torch.manual_seed(0)
    p_ours = torch.nn.Parameter(torch.randn(10))
    p_ref = torch.nn.Parameter(p_ours.detach().clone())
    cfg = _cfg('Adam')
    opt = Optimizer(cfg, [p_ours])
    ref = torch.optim.Adam([p_ref], lr=cfg.initial_learning_rate,
                           betas=(cfg.beta1, cfg.beta2), eps=cfg.epsilon)
    for step in range(5):
        g = torch.randn(10)
        p_ours.grad = g.clone()
        p_ref.grad = g.clone()
        opt.step()
        ref.step()
        assert torch.allclose(p_ours, p_ref, atol=1e-6), step


def test_global_norm_clip():
    p = torch.nn.Parameter(torch.zeros(4))
    cfg = _cfg('SGD', clip=1.0)
    opt = Optimizer(cfg, [p])
    p.grad = torch.full((4,), 10.0)  # norm 20 -> scaled to 1
    opt.step()
    # SGD: p -= lr * clipped_grad; clipped grad = g/20
    expected = -cfg.initial_learning_rate * (10.0 / 20.0)
    assert torch.allclose(p.data, torch.full((4,), expected), atol=1e-6)


def test_momentum_nesterov_runs():
    p = torch.nn.Parameter(torch.ones(3))
    cfg = _cfg('Momentum')
    cfg.momentum = 0.9
    opt = Optimizer(cfg, [p])
    for _ in range(3):
        p.grad = torch.ones(3)
        opt.step()
    assert torch.isfinite(p).all()


def test_rmsprop_runs():
    p = torch.nn.Parameter(torch.ones(3))
    cfg = _cfg('RMSProp')
    opt = Optimizer(cfg, [p])
    p.grad = torch.ones(3)
    opt.step()
    assert torch.isfinite(p).all()


def test_lr_decay_staircase():
    cfg = _cfg()
    cfg.learning_rate_decay_factor = 0.5
    cfg.num_steps_per_decay = 10
    opt = Optimizer(cfg, [torch.nn.Parameter(torch.ones(1))])
    opt.step_count = 5
    assert abs(opt.learning_rate() - 0.01) < 1e-12
    opt.step_count = 10
    assert abs(opt.learning_rate() - 0.005) < 1e-12


def test_optimizer_state_roundtrip():
    p = torch.nn.Parameter(torch.randn(5))
    cfg = _cfg('Adam')
    opt = Optimizer(cfg, [p])
    p.grad = torch.randn(5)
    opt.step()
    arrays = {k: v.numpy() for k, v in opt.state_arrays().items()}

    p2 = torch.nn.Parameter(torch.randn(5))
    opt2 = Optimizer(cfg, [p2])
    opt2.load_state_arrays(arrays)
    assert opt2.step_count == 1
    assert torch.allclose(opt2.state[p2]['m'], opt.state[p]['m'])
