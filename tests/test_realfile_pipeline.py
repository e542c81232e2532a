"""The real-file data path (the default, non-synthetic one): COCO captions
JSON + JPEG files on disk -> vocabulary/caption caches -> training ->
eval with overlay artifacts.  Everything the reference's file-backed flow
does, minus the network download."""

import json
import os

import numpy as np
import pytest
from PIL import Image

from sat_amd.data.dataset import (build_vocabulary, prepare_eval_data,
                                  prepare_train_data)
from sat_amd.models.base_model import BaseModel

CAPS = ['a red dog sitting on a table.', 'a small cat on a beach.',
        'a young man riding a horse.', 'a large bus on a street.',
        'a white bird on a bench.', 'a black boat near a field.']


def _make_dataset(root, n, prefix, start_id=1):
    img_dir = os.path.join(root, 'images')
    os.makedirs(img_dir, exist_ok=True)
    images, anns = [], []
    rng = np.random.RandomState(0)
    for i in range(n):
        iid = start_id + i
        name = '%s_%06d.jpg' % (prefix, iid)
        arr = rng.randint(0, 255, (48, 64, 3), dtype=np.uint8)
        Image.fromarray(arr).save(os.path.join(img_dir, name))
        images.append({'id': iid, 'file_name': name})
        anns.append({'id': iid, 'image_id': iid,
                     'caption': CAPS[i % len(CAPS)]})
    caps_file = os.path.join(root, 'captions.json')
    with open(caps_file, 'w') as f:
        json.dump({'images': images, 'annotations': anns}, f)
    return img_dir, caps_file


def test_real_file_train_and_eval(tiny_config, tmp_path):
    cfg = tiny_config
    cfg.synthetic_data = False
    cfg.vocabulary_size = 100
    train_dir, train_caps = _make_dataset(str(tmp_path / 'train'), 6, 'tr')
    val_dir, val_caps = _make_dataset(str(tmp_path / 'val'), 3, 'va',
                                      start_id=100)
    cfg.train_image_dir = train_dir
    cfg.train_caption_file = train_caps
    cfg.eval_image_dir = val_dir
    cfg.eval_caption_file = val_caps
    cfg.temp_annotation_file = str(tmp_path / 'anns.csv')
    cfg.temp_data_file = str(tmp_path / 'data.npy')
    cfg.vocabulary_file = str(tmp_path / 'vocabulary.csv')
    cfg.max_train_ann_num = 6
    cfg.max_eval_ann_num = 3
    cfg.num_epochs = 1
    cfg.batch_size = 2

    data = prepare_train_data(cfg)
    assert os.path.exists(cfg.vocabulary_file)
    assert os.path.exists(cfg.temp_annotation_file)
    assert os.path.exists(cfg.temp_data_file)
    m = BaseModel(cfg)
    m.train(data)
    assert m.global_step == data.num_batches

    # second prepare run consumes the caches
    data2 = prepare_train_data(cfg)
    assert data2.count == data.count

    cfg.phase = 'eval'
    cfg.batch_size = 1
    cfg.save_eval_result_as_image = True
    coco, ds, vocab = prepare_eval_data(cfg)
    m2 = BaseModel(cfg)
    m2.load()
    scores = m2.eval(coco, ds, vocab)
    assert 'Bleu_1' in scores
    with open(cfg.eval_result_file) as f:
        results = json.load(f)
    assert len(results) == 3
    overlays = [f for f in os.listdir(cfg.eval_result_dir)
                if f.endswith('_result.jpg')]
    assert len(overlays) == 3


def test_build_vocabulary_from_files(tiny_config, tmp_path):
    cfg = tiny_config
    cfg.synthetic_data = False
    _, caps = _make_dataset(str(tmp_path / 't'), 4, 'x')
    cfg.train_caption_file = caps
    cfg.vocabulary_file = str(tmp_path / 'v.csv')
    cfg.max_train_ann_num = 4
    v = build_vocabulary(cfg)
    assert v.words[0] == '<start>'
    assert os.path.exists(cfg.vocabulary_file)


def test_real_file_test_phase(tiny_config, tmp_path):
    """--phase=test on arbitrary JPEGs (reference base_model.py:119-161)."""
    from sat_amd.data.dataset import prepare_test_data
    cfg = tiny_config
    cfg.synthetic_data = False
    img_dir, caps = _make_dataset(str(tmp_path / 'tr'), 4, 'y')
    cfg.train_caption_file = caps
    cfg.vocabulary_file = str(tmp_path / 'v.csv')
    cfg.max_train_ann_num = 4
    cfg.test_image_dir = img_dir
    cfg.test_result_dir = str(tmp_path / 'res') + '/'
    cfg.test_result_file = str(tmp_path / 'res.csv')
    cfg.batch_size = 2
    cfg.phase = 'test'
    ds, vocab = prepare_test_data(cfg)
    assert ds.count == 4
    m = BaseModel(cfg)
    df = m.test(ds, vocab)
    assert os.path.exists(cfg.test_result_file)
    assert len(df) == 4
    overlays = [f for f in os.listdir(cfg.test_result_dir)
                if f.endswith('_result.jpg')]
    assert len(overlays) == 4
