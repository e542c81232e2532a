"""The driver depends on bench.py's CLI + JSON contract; pin it."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, 'bench.py', '--steps', '1', '--warmup', '0',
         '--batch', '1'],
        capture_output=True, text=True, cwd=ROOT, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
                'ms_per_step', 'higher_is_better', 'scaling',
                'vs_baseline', 'dtype', 'data', 'config'):
        assert key in d, key
    assert d['metric'] == 'training images/sec'
    assert d['higher_is_better'] is True
    assert d['scaling'] == 'weak'
    assert d['data'] == 'synthetic'
    assert d['value'] > 0
    assert {'model', 'global_batch', 'seq_len',
            'parallelism'} <= set(d['config'])


def test_graft_entry_has_build_and_smoke():
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        'graft_entry', os.path.join(ROOT, '__graft_entry__.py'))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    assert callable(mod.build) and callable(mod.smoke)
