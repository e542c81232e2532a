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
