"""Checkpoint save/load/trim — `.npy`-dict layout (parity with reference
`base_model.py:242-297` + `data/models/trim_model.py`).

Layout: one `numpy.save`d dict {variable_name: ndarray} holding every model
parameter/buffer, every optimizer slot (named 'optimizer/...'), and
'global_step'; written to `<save_dir>/<global_step>.npy`, with the Config
object pickled alongside as `config.pickle` stamped with global_step
(base_model.py:245-254).  Loading matches per-variable BY NAME with silent
skip on miss, printing the match count (base_model.py:272-278), and can
discover the latest checkpoint through config.pickle (base_model.py:260-269).

`load_cnn` accepts a Caffe-style dict {scope: {param_name: ndarray}} for
pretrained encoders (base_model.py:280-297), accepting both TF [kh,kw,in,out]
and torch [out,in,kh,kw] conv kernel layouts.
"""

import copy
import os
import pickle

import numpy as np
import torch


def save(model, optimizer, config, global_step):
    """Dump all variables + optimizer slots to <save_dir>/<step>.npy."""
    os.makedirs(config.save_dir, exist_ok=True)
    arrays = {}
    for name, t in model.state_dict().items():
        arrays[name] = t.detach().cpu().float().numpy()
    if optimizer is not None:
        for name, t in optimizer.state_arrays().items():
            arrays[name] = t.detach().cpu().float().numpy()
    arrays['global_step'] = np.array(float(global_step))

    path = os.path.join(config.save_dir, '%d.npy' % global_step)
    print(' Saving the model to %s...' % path)
    np.save(path, arrays)

    cfg = copy.copy(config)
    cfg.global_step = global_step
    with open(os.path.join(config.save_dir, 'config.pickle'), 'wb') as f:
        pickle.dump(cfg, f)
    print('Model saved.')
    return path


def _discover_latest(config):
    info_path = os.path.join(config.save_dir, 'config.pickle')
    with open(info_path, 'rb') as f:
        cfg = pickle.load(f)
    return os.path.join(config.save_dir, '%d.npy' % cfg.global_step)


def load(model, optimizer, config, model_file=None):
    """Name-matched restore; returns global_step."""
    path = model_file or _discover_latest(config)
    print('Loading the model from %s...' % path)
    arrays = np.load(path, allow_pickle=True).item()

    state = model.state_dict()
    count = 0
    with torch.no_grad():
        for name, t in state.items():
            if name in arrays:
                arr = np.asarray(arrays[name])
                if tuple(arr.shape) == tuple(t.shape):
                    t.copy_(torch.as_tensor(arr).to(t.device, t.dtype))
                    count += 1
    if optimizer is not None:
        optimizer.load_state_arrays(arrays)
    global_step = int(float(arrays.get('global_step', 0)))
    print('%d tensors loaded.' % count)
    # Name-matching silently skips misses (reference base_model.py:272-278
    # semantics) — but a checkpoint from a DIFFERENT naming scheme (e.g. a
    # reference TF .npy with names like 'word_embedding:0') matches zero
    # tensors and would otherwise be indistinguishable from a good load.
    # Such files need a name-translation pass first (tools/, docs/).
    if count == 0 and len(state) > 0:
        raise RuntimeError(
            'checkpoint %s matched 0 of %d model tensors — wrong file or '
            'incompatible variable naming (reference TF checkpoints need '
            'name translation before loading)' % (path, len(state)))
    if count < len(state) // 2:
        print('WARNING: only %d of %d model tensors matched %s — partial '
              'restore' % (count, len(state), path))
    return global_step


def load_cnn(model, data_path, verbose=True):
    """Load a pretrained CNN from a Caffe-style {scope: {param: array}} dict
    (or a flat {name: array} dict) into model.cnn, matching by scope name."""
    print('Loading the CNN from %s...' % data_path)
    data = np.load(data_path, allow_pickle=True, encoding='latin1').item()
    data = {_translate_scope(k): v for k, v in data.items()}
    cnn_state = dict(model.cnn.named_parameters())
    cnn_state.update(dict(model.cnn.named_buffers()))
    count = 0
    _PARAM_MAP = {'weights': 'weight', 'kernel': 'weight',
                  'biases': 'bias', 'bias': 'bias',
                  'mean': 'bn.running_mean', 'variance': 'bn.running_var',
                  'scale': 'bn.weight', 'offset': 'bn.bias',
                  'gamma': 'bn.weight', 'beta': 'bn.bias'}
    with torch.no_grad():
        for scope, params in data.items():
            if not isinstance(params, dict):
                continue
            for pname, arr in params.items():
                tgt = '%s.%s' % (scope, _PARAM_MAP.get(pname, pname))
                if tgt not in cnn_state:
                    continue
                t = cnn_state[tgt]
                arr = np.asarray(arr)
                if arr.ndim == 4 and tuple(arr.shape) != tuple(t.shape):
                    arr = arr.transpose(3, 2, 0, 1)  # TF HWIO -> torch OIHW
                if tuple(arr.shape) == tuple(t.shape):
                    t.copy_(torch.as_tensor(arr).to(t.device, t.dtype))
                    count += 1
    print('%d tensors loaded.' % count)
    return count


_BRANCH = {'branch2a': 'conv_a', 'branch2b': 'conv_b',
           'branch2c': 'conv_c', 'branch1': 'shortcut'}


def _translate_scope(scope):
    """Map Caffe-style ResNet scope names (res2a_branch2a, bn2a_branch1,
    ...) onto this repo's module paths (res2a.conv_a, res2a.shortcut_bn).
    VGG scopes (conv1_1, ...) pass through unchanged."""
    for pre, kind in (('res', ''), ('bn', 'bn')):
        if scope.startswith(pre) and '_branch' in scope:
            block, branch = scope[len(pre):].split('_', 1)
            if branch in _BRANCH:
                leaf = _BRANCH[branch]
                if kind == 'bn':
                    leaf = ('shortcut_bn' if leaf == 'shortcut'
                            else 'bn' + leaf[-2:])
                return 'res%s.%s' % (block, leaf)
    return scope


def trim(model_file, out_file=None):
    """Strip optimizer slots from a checkpoint (reference trim_model.py)."""
    arrays = np.load(model_file, allow_pickle=True).item()
    trimmed = {k: v for k, v in arrays.items() if 'optimizer' not in k}
    out_file = out_file or model_file
    np.save(out_file, trimmed)
    return len(arrays) - len(trimmed)
