"""End-to-end pipeline tests: train -> checkpoint -> eval -> test on the
synthetic corpus (the BASELINE Config #1 plumbing slice), all CPU."""

import copy
import json
import os

import pytest

from sat_amd.data.dataset import (prepare_eval_data, prepare_test_data,
                                  prepare_train_data)
from sat_amd.models.base_model import BaseModel


def test_train_eval_test_pipeline(tiny_config):
    cfg = tiny_config

    data = prepare_train_data(cfg)
    model = BaseModel(cfg)
    model.train(data)
    assert model.global_step == data.num_batches
    ckpt_path = os.path.join(cfg.save_dir,
                             '%d.npy' % model.global_step)
    assert os.path.exists(ckpt_path)
    assert os.path.exists(os.path.join(cfg.save_dir, 'config.pickle'))
    assert os.path.exists(os.path.join(cfg.summary_dir, 'events.jsonl'))

    # eval phase (batch forced to 1 like main.py does)
    cfg_eval = copy.copy(cfg)
    cfg_eval.phase = 'eval'
    cfg_eval.batch_size = 1
    coco, ds, vocab = prepare_eval_data(cfg_eval)
    m2 = BaseModel(cfg_eval)
    m2.load()
    scores = m2.eval(coco, ds, vocab)
    assert set(scores) >= {'Bleu_1', 'Bleu_4', 'METEOR', 'ROUGE_L',
                           'CIDEr'}
    with open(cfg_eval.eval_result_file) as f:
        results = json.load(f)
    assert len(results) == len(ds.image_ids)
    assert all('image_id' in r and 'caption' in r for r in results)

    # test phase
    cfg_test = copy.copy(cfg)
    cfg_test.phase = 'test'
    ds, vocab = prepare_test_data(cfg_test)
    m3 = BaseModel(cfg_test)
    m3.load()
    df = m3.test(ds, vocab)
    assert os.path.exists(cfg_test.test_result_file)
    assert len(df) == len(ds.image_ids)


def test_resume_from_checkpoint(tiny_config):
    cfg = tiny_config
    data = prepare_train_data(cfg)
    m = BaseModel(cfg)
    m.train(data)
    step = m.global_step

    m2 = BaseModel(cfg)
    m2.load()  # discover via config.pickle
    assert m2.global_step == step
    data.reset()
    m2.train(data)
    assert m2.global_step == 2 * step


def test_cli_flags_parity():
    from main import build_parser
    p = build_parser()
    args = p.parse_args(['--phase=eval', '--load', '--model_file=x.npy',
                         '--load_cnn', '--cnn_model_file=c.npy',
                         '--train_cnn', '--beam_size=5'])
    assert args.phase == 'eval' and args.load and args.train_cnn
    assert args.model_file == 'x.npy'
    assert args.cnn_model_file == 'c.npy'
    assert args.beam_size == 5


def test_trim_model_cli(tiny_config, tmp_path):
    """tools/trim_model.py end-to-end (reference trim_model.py parity)."""
    import subprocess
    import sys
    import numpy as np
    from sat_amd.models.base_model import BaseModel
    cfg = tiny_config
    m = BaseModel(cfg)
    path = m.save()
    out = str(tmp_path / 'trimmed.npy')
    r = subprocess.run(
        [sys.executable, 'tools/trim_model.py', path, out],
        capture_output=True, text=True,
        cwd=os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr
    arrays = np.load(out, allow_pickle=True).item()
    assert not any('optimizer' in k for k in arrays)


def test_eval_sh_exists_and_executable():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    p = os.path.join(root, 'eval.sh')
    assert os.path.exists(p)
    assert os.access(p, os.X_OK)


def test_variable_summaries_written(tiny_config):
    """Per-variable stats land in the event file at checkpoint cadence."""
    import json as _json
    from sat_amd.data.dataset import prepare_train_data as prep
    from sat_amd.models.base_model import BaseModel
    cfg = tiny_config
    cfg.save_period = 2
    data = prep(cfg)
    m = BaseModel(cfg)
    m.train(data)
    path = os.path.join(cfg.summary_dir, 'events.jsonl')
    recs = [_json.loads(l) for l in open(path)]
    assert any(any(k.endswith('/mean') for k in r) for r in recs)
    assert any('total_loss' in r for r in recs)


def test_config_env_overrides(monkeypatch):
    import json as _json
    from config import Config
    monkeypatch.setenv('SAT_CONFIG_OVERRIDES',
                       _json.dumps({'batch_size': 7, 'cnn': 'resnet50'}))
    cfg = Config()
    assert cfg.batch_size == 7 and cfg.cnn == 'resnet50'


def test_config_knob_names_match_reference():
    """Every reference config.py knob must exist under the same name."""
    from config import Config
    cfg = Config()
    knobs = [
        'cnn', 'max_caption_length', 'dim_embedding', 'num_lstm_units',
        'num_initalize_layers', 'dim_initalize_layer', 'num_attend_layers',
        'dim_attend_layer', 'num_decode_layers', 'dim_decode_layer',
        'fc_kernel_initializer_scale', 'fc_kernel_regularizer_scale',
        'fc_activity_regularizer_scale', 'conv_kernel_regularizer_scale',
        'conv_activity_regularizer_scale', 'fc_drop_rate',
        'lstm_drop_rate', 'attention_loss_factor', 'num_epochs',
        'batch_size', 'optimizer', 'initial_learning_rate',
        'learning_rate_decay_factor', 'num_steps_per_decay',
        'clip_gradients', 'momentum', 'use_nesterov', 'decay', 'centered',
        'beta1', 'beta2', 'epsilon', 'path_to_local_logs',
        'root_path_to_local_data', 'local_repo', 'cloud_user_repo',
        'cloud_path_to_data', 'save_period', 'save_dir', 'summary_dir',
        'max_train_ann_num', 'max_eval_ann_num', 'vocabulary_file',
        'vocabulary_size', 'train_image_dir', 'train_caption_file',
        'temp_annotation_file', 'temp_data_file', 'eval_image_dir',
        'eval_caption_file', 'eval_result_dir', 'eval_result_file',
        'save_eval_result_as_image', 'test_image_dir', 'test_result_dir',
        'test_result_file',
    ]
    missing = [k for k in knobs if not hasattr(cfg, k)]
    assert not missing, missing
