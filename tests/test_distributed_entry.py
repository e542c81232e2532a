"""End-to-end test of the distributed entry script: 2 ranks over gloo on
CPU via torchrun, tiny synthetic corpus — the same code path the 8-GPU
RCCL launch uses (sat_amd.parallel.launch resolves the backend)."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_main_distributed_two_ranks(tmp_path):
    overrides = {
        'synthetic_data': True,
        'synthetic_num_images': 8,
        'batch_size': 2,
        'num_epochs': 1,
        'max_train_ann_num': 8,
        'vocabulary_size': 60,
        'dim_embedding': 16,
        'num_lstm_units': 16,
        'dim_initalize_layer': 16,
        'dim_attend_layer': 16,
        'dim_decode_layer': 16,
        'save_period': 10 ** 9,
        'save_dir': str(tmp_path / 'models') + '/',
        'summary_dir': str(tmp_path / 'summary') + '/',
        'device': 'cpu',
    }
    env = dict(os.environ)
    env['SAT_CONFIG_OVERRIDES'] = json.dumps(overrides)
    r = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29617', 'main_distributed.py',
         '--phase=train', '--synthetic', '--device', 'cpu'],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=600)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    # chief-only checkpoint written, with each rank's 2-image shard
    # giving num_batches = 1 -> global_step 1
    files = os.listdir(str(tmp_path / 'models'))
    assert any(f.endswith('.npy') for f in files)
    assert 'config.pickle' in files
    assert os.path.exists(str(tmp_path / 'summary') + '/events.jsonl')
