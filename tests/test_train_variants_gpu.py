"""GPU coverage for the non-flagship training variants: --train_cnn
(end-to-end CNN+RNN) and the ResNet50 encoder."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _batch(V, B, T=20):
    torch.manual_seed(0)
    images = torch.randn(B, 3, 224, 224, device='cuda') * 50.0
    sentences = torch.randint(1, V, (B, T), device='cuda')
    masks = torch.ones(B, T, device='cuda')
    return images, sentences, masks


def test_train_cnn_end_to_end(tiny_config):
    """--train_cnn: conv weights must receive gradients and update."""
    from sat_amd.models.base_model import BaseModel
    cfg = tiny_config
    cfg.device = 'cuda'
    cfg.train_cnn = True
    cfg.vocabulary_size = 500
    cfg.dim_embedding = 512
    cfg.num_lstm_units = 512
    cfg.dim_initalize_layer = 512
    cfg.dim_attend_layer = 512
    cfg.dim_decode_layer = 1024
    cfg.use_hip_graph = False  # keep the variant test simple/eager
    torch.manual_seed(cfg.seed)
    m = BaseModel(cfg)
    conv_w = m.model.cnn.conv1_1.weight
    assert conv_w.requires_grad
    before = conv_w.detach().clone()
    out = m.train_step(*_batch(500, 2))
    torch.cuda.synchronize()
    assert torch.isfinite(out['total_loss'])
    assert conv_w.grad is not None or not torch.equal(
        before, conv_w.detach())
    # optimizer covers CNN params too
    assert any(p is conv_w for p in m.optimizer.params)


def test_resnet50_training_step(tiny_config):
    from sat_amd.models.base_model import BaseModel
    cfg = tiny_config
    cfg.device = 'cuda'
    cfg.cnn = 'resnet50'
    cfg.vocabulary_size = 500
    cfg.dim_embedding = 512
    cfg.num_lstm_units = 512
    cfg.dim_initalize_layer = 512
    cfg.dim_attend_layer = 512
    cfg.dim_decode_layer = 1024
    torch.manual_seed(cfg.seed)
    m = BaseModel(cfg)
    assert m.model.num_ctx == 49 and m.model.dim_ctx == 2048
    losses = [m.train_step(*_batch(500, 2))['total_loss'].item()
              for _ in range(6)]
    assert all(x == x for x in losses)
    assert losses[-1] < losses[0]


def test_resnet50_beam_search(tiny_config):
    from sat_amd.data.dataset import prepare_eval_data
    from sat_amd.models.base_model import BaseModel
    cfg = tiny_config
    cfg.device = 'cuda'
    cfg.cnn = 'resnet50'
    cfg.phase = 'eval'
    cfg.batch_size = 1
    coco, data, vocab = prepare_eval_data(cfg)
    m = BaseModel(cfg)
    files = data.next_batch()
    results = m.beam_search(files, vocab)
    assert len(results) == 1 and len(results[0]) >= 1
