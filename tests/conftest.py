import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X (or any ROCm GPU); run with "
                   "`pytest -m gpu` on a GPU box")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_config(tmp_path):
    """BASELINE Config #1-shaped tiny config on synthetic data."""
    from config import Config
    cfg = Config()
    cfg.phase = 'train'
    cfg.train_cnn = False
    cfg.beam_size = 2
    cfg.synthetic_data = True
    cfg.synthetic_num_images = 8
    cfg.batch_size = 2
    cfg.num_epochs = 1
    cfg.max_train_ann_num = 8
    cfg.max_eval_ann_num = 4
    cfg.vocabulary_size = 100
    cfg.dim_embedding = 32
    cfg.num_lstm_units = 32
    cfg.dim_initalize_layer = 32
    cfg.dim_attend_layer = 32
    cfg.dim_decode_layer = 48
    cfg.save_period = 1000
    cfg.save_dir = str(tmp_path / 'models') + '/'
    cfg.summary_dir = str(tmp_path / 'summary') + '/'
    cfg.eval_result_dir = str(tmp_path / 'eval') + '/'
    cfg.eval_result_file = str(tmp_path / 'eval_results.json')
    cfg.test_result_dir = str(tmp_path / 'test') + '/'
    cfg.test_result_file = str(tmp_path / 'test_results.csv')
    cfg.vocabulary_file = str(tmp_path / 'vocabulary.csv')
    cfg.save_eval_result_as_image = False
    cfg.device = 'cpu'
    return cfg
