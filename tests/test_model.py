import numpy as np
import torch

from sat_amd.models.base_model import BaseModel
from sat_amd.models.caption_generator import CaptionGenerator
from sat_amd.utils import checkpoint as ckpt


def _batch(cfg, B=2):
    torch.manual_seed(0)
    images = torch.randn(B, 3, 224, 224)
    T = cfg.max_caption_length
    sentences = torch.randint(0, cfg.vocabulary_size, (B, T))
    masks = torch.zeros(B, T)
    masks[:, :5] = 1.0
    return images, sentences, masks


def test_forward_losses(tiny_config):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    model = CaptionGenerator(cfg)
    out = model(*_batch(cfg))
    for k in ('total_loss', 'cross_entropy_loss', 'attention_loss',
              'reg_loss', 'accuracy'):
        assert torch.isfinite(out[k]), k
    assert out['total_loss'].item() >= out['cross_entropy_loss'].item()
    assert 0.0 <= out['accuracy'].item() <= 1.0
    assert out['attentions'].shape == (2, model.num_ctx)


def test_backward_populates_rnn_grads_only(tiny_config):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    model = CaptionGenerator(cfg)
    out = model(*_batch(cfg))
    out['total_loss'].backward()
    dec_grads = [p.grad for p in model.decoder.parameters()]
    assert all(g is not None for g in dec_grads)
    assert all(torch.isfinite(g).all() for g in dec_grads)
    # frozen CNN: no grads
    assert all(p.grad is None for p in model.cnn.parameters())


def test_encode_decode_step_shapes(tiny_config):
    cfg = tiny_config
    cfg.phase = 'eval'
    cfg.vocabulary_size = 50
    model = CaptionGenerator(cfg)
    model.eval()
    images = torch.randn(2, 3, 224, 224)
    ctx, mem, out = model.encode(images)
    assert ctx.shape == (2, 196, 512)
    assert mem.shape == (2, cfg.num_lstm_units)
    word = torch.zeros(2, dtype=torch.int64)
    mem2, out2, probs = model.decode_step(ctx, word, mem, out)
    assert probs.shape == (2, 50)
    assert torch.allclose(probs.sum(1), torch.ones(2), atol=1e-5)


def test_checkpoint_roundtrip(tiny_config):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    m = BaseModel(cfg)
    m.global_step = 7
    path = m.save()
    arrays = np.load(path, allow_pickle=True).item()
    assert 'global_step' in arrays
    assert any(k.startswith('optimizer/') for k in arrays)

    m2 = BaseModel(cfg)
    step = m2.load(path)
    assert step == 7
    for (n1, p1), (n2, p2) in zip(m.model.state_dict().items(),
                                  m2.model.state_dict().items()):
        assert n1 == n2
        assert torch.allclose(p1.float(), p2.float(), atol=1e-6), n1


def test_checkpoint_trim(tiny_config, tmp_path):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    m = BaseModel(cfg)
    path = m.save()
    out = str(tmp_path / 'trimmed.npy')
    removed = ckpt.trim(path, out)
    assert removed > 0
    arrays = np.load(out, allow_pickle=True).item()
    assert not any('optimizer' in k for k in arrays)


def test_load_cnn_caffe_style(tiny_config, tmp_path):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    m = BaseModel(cfg)
    w = np.random.randn(64, 3, 3, 3).astype(np.float32)
    b = np.random.randn(64).astype(np.float32)
    data = {'conv1_1': {'weights': w, 'biases': b}}
    f = str(tmp_path / 'cnn.npy')
    np.save(f, data)
    count = m.load_cnn(f)
    assert count == 2
    got = dict(m.model.cnn.named_parameters())['conv1_1.weight']
    assert torch.allclose(got, torch.from_numpy(w))


def test_load_cnn_tf_layout(tiny_config, tmp_path):
    cfg = tiny_config
    cfg.vocabulary_size = 50
    m = BaseModel(cfg)
    w_tf = np.random.randn(3, 3, 3, 64).astype(np.float32)  # HWIO
    data = {'conv1_1': {'kernel': w_tf}}
    f = str(tmp_path / 'cnn_tf.npy')
    np.save(f, data)
    assert m.load_cnn(f) == 1
    got = dict(m.model.cnn.named_parameters())['conv1_1.weight']
    assert torch.allclose(got, torch.from_numpy(
        w_tf.transpose(3, 2, 0, 1)))


def test_beam_search_semantics(tiny_config):
    cfg = tiny_config
    cfg.phase = 'eval'
    from sat_amd.data.synthetic import prepare_eval_data
    coco, data, vocab = prepare_eval_data(cfg)
    m = BaseModel(cfg)
    files = data.next_batch()
    results = m.beam_search(files, vocab)
    assert len(results) == len(files)
    for beams in results:
        assert 1 <= len(beams) <= cfg.beam_size
        scores = [b.score for b in beams]
        assert scores == sorted(scores, reverse=True)
        for b in beams:
            assert len(b.sentence) <= cfg.max_caption_length
            assert all(isinstance(w, int) for w in b.sentence)


def test_beam_search_deterministic(tiny_config):
    cfg = tiny_config
    cfg.phase = 'eval'
    from sat_amd.data.synthetic import prepare_eval_data
    coco, data, vocab = prepare_eval_data(cfg)
    m = BaseModel(cfg)
    files = data.next_batch()
    r1 = m.beam_search(files, vocab)
    r2 = m.beam_search(files, vocab)
    assert [b.sentence for b in r1[0]] == [b.sentence for b in r2[0]]


def test_load_cnn_caffe_resnet_scopes(tiny_config, tmp_path):
    """Caffe-style ResNet scope names translate onto our module paths."""
    cfg = tiny_config
    cfg.cnn = 'resnet50'
    cfg.vocabulary_size = 50
    m = BaseModel(cfg)
    w = np.random.randn(64, 64, 3, 3).astype(np.float32)  # res2a conv_b
    gamma = np.random.randn(64).astype(np.float32)
    data = {'res2a_branch2b': {'weights': w},
            'bn2a_branch2b': {'gamma': gamma}}
    f = str(tmp_path / 'resnet.npy')
    np.save(f, data)
    count = m.load_cnn(f)
    assert count == 2
    got = dict(m.model.cnn.named_parameters())['res2a.conv_b.weight']
    assert torch.allclose(got, torch.from_numpy(w))


def test_coco_cat_stubs():
    from sat_amd.data.coco import COCO
    c = COCO()
    c.dataset = {'images': [], 'annotations': [],
                 'categories': [{'id': 5, 'name': 'dog'}]}
    c.createIndex()
    assert c.getCatIds() == [5]
    assert c.loadCats(5)[0]['name'] == 'dog'


def test_single_layer_mlp_variants(tiny_config):
    """num_initalize/attend/decode_layers == 1 (reference model.py
    supports both); forward + backward must stay finite."""
    cfg = tiny_config
    cfg.vocabulary_size = 50
    cfg.num_initalize_layers = 1
    cfg.num_attend_layers = 1
    cfg.num_decode_layers = 1
    model = CaptionGenerator(cfg)
    out = model(*_batch(cfg))
    assert torch.isfinite(out['total_loss'])
    out['total_loss'].backward()
    assert all(torch.isfinite(p.grad).all()
               for p in model.decoder.parameters() if p.grad is not None)


def test_activity_regularizers_apply():
    """fc/conv activity regularizers (reference nn.py:23-27,39-43) are
    L1-of-activations terms in reg_loss — not silent dead knobs."""
    import torch
    from config import Config
    from sat_amd.models.nn import NN, Dense

    cfg = Config()
    cfg.phase = 'train'
    cfg.train_cnn = False
    cfg.fc_activity_regularizer_scale = 0.01
    pol = NN(cfg)
    d = Dense(pol, 4, 3, 'tanh')
    x = torch.randn(2, 4)
    y = d(x)
    expected = cfg.fc_kernel_regularizer_scale * 0.5 \
        * float((d.weight ** 2).sum()) \
        + 0.01 * float(y.abs().sum())
    got = float(pol.reg_loss())
    assert abs(got - expected) < 1e-5
    # cleared after reg_loss
    assert pol._act_losses == []
    # activation-free layers carry no activity term (reference nn.py:92-95)
    d2 = Dense(pol, 4, 3, None)
    d2(x)
    assert pol._act_losses == []


def test_activity_regularizer_reaches_total_loss():
    import torch
    from config import Config
    from sat_amd.models.caption_generator import CaptionGenerator

    def tiny(scale):
        cfg = Config()
        cfg.phase = 'train'
        cfg.train_cnn = False
        cfg.fc_activity_regularizer_scale = scale
        cfg.vocabulary_size = 30
        cfg.dim_embedding = 16
        cfg.num_lstm_units = 16
        cfg.dim_initalize_layer = 16
        cfg.dim_attend_layer = 16
        cfg.dim_decode_layer = 16
        cfg.max_caption_length = 4
        torch.manual_seed(0)
        m = CaptionGenerator(cfg)
        torch.manual_seed(1)
        img = torch.randn(1, 3, 224, 224)
        s = torch.randint(0, 30, (1, 4))
        mk = torch.ones(1, 4)
        return float(m(img, s, mk)['reg_loss'])

    assert tiny(0.05) > tiny(0.0)


def test_device_beam_matches_host_beam():
    """The device-resident beam scorer must produce the same captions
    and scores as the host-heap reference path."""
    import torch
    from config import Config
    from sat_amd.models.base_model import BaseModel
    from sat_amd.data.vocabulary import Vocabulary

    cfg = Config()
    cfg.phase = 'eval'
    cfg.train_cnn = False
    cfg.synthetic_data = True
    cfg.device = 'cpu'
    cfg.beam_size = 3
    cfg.vocabulary_size = 40
    cfg.dim_embedding = 16
    cfg.num_lstm_units = 16
    cfg.dim_initalize_layer = 16
    cfg.dim_attend_layer = 16
    cfg.dim_decode_layer = 16
    cfg.max_caption_length = 8

    vocab = Vocabulary(40)
    vocab.build(['a man rides a horse down the street .',
                 'a dog sits on the beach sand .',
                 'the cat sleeps near a window ledge .'])

    torch.manual_seed(3)
    m = BaseModel(cfg)
    files = ['synthetic://0', 'synthetic://1']

    host = m.beam_search_host(files, vocab)
    dev = m.beam_search_device(files, vocab)

    for k in range(len(files)):
        hs = [(h.sentence, h.score) for h in host[k]]
        ds = [(list(map(int, d.sentence)), d.score) for d in dev[k]]
        assert len(hs) == len(ds), (k, len(hs), len(ds))
        for (s1, p1), (s2, p2) in zip(hs, ds):
            assert s1 == s2, (k, s1, s2)
            assert abs(p1 - p2) < 1e-6 * max(abs(p1), 1e-30), (p1, p2)


def test_chained_pad_tag_unpads_on_fallback():
    """A tensor tagged `_sat_pad` by a padded-emit conv must be unpadded
    before any consumer that cannot use it (library conv, non-accepting
    layer) — the handshake degrades safely."""
    import torch
    from config import Config
    from sat_amd.models.nn import NN, Conv2d

    cfg = Config()
    cfg.phase = 'eval'
    pol = NN(cfg)
    conv = Conv2d(pol, 8, 8, 3, 1, 'relu')
    torch.manual_seed(21)
    x = torch.randn(1, 8, 6, 6)
    ref = conv(x)

    xp = torch.zeros(1, 8, 8, 8)
    xp[:, :, 1:7, 1:7] = x
    xp._sat_pad = (6, 6)
    got = conv(xp)  # CPU conv cannot consume padded input -> must unpad
    assert got.shape == ref.shape
    assert torch.allclose(got, ref, atol=1e-5)
