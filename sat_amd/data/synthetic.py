"""Synthetic COCO-shaped data (no network, no files).

This environment cannot download COCO (the reference auto-downloads images per
run, `utils/coco/coco.py:292-314`), so every pipeline phase can instead run on
a fabricated corpus with exactly the COCO shape the reference consumes:
captions JSON {'images': [...], 'annotations': [...]}, 224x224 RGB images,
5-captions-per-image val structure.  Deterministic in config.seed.

Image "files" use the `synthetic://<image_id>` scheme; SyntheticImageLoader
turns them into deterministic random mean-subtracted float32 arrays of the
right shape, so the whole train/eval/test stack exercises identical code
paths to the file-backed one.
"""

import numpy as np

from .coco import COCO
from .dataset import DataSet, _tensorize_captions
from .vocabulary import Vocabulary

_NOUNS = ['man', 'woman', 'dog', 'cat', 'bus', 'car', 'plate', 'table',
          'street', 'beach', 'horse', 'train', 'pizza', 'kite', 'bench',
          'boy', 'girl', 'bird', 'boat', 'field']
_VERBS = ['riding', 'holding', 'sitting', 'standing', 'walking', 'eating',
          'playing', 'looking', 'jumping', 'running']
_ADJS = ['red', 'small', 'large', 'young', 'old', 'white', 'black', 'green']
_PREPS = ['on', 'near', 'with', 'beside', 'under', 'over']


def _make_caption(rng):
    words = ['a', rng.choice(_ADJS), rng.choice(_NOUNS), rng.choice(_VERBS),
             rng.choice(_PREPS), 'a', rng.choice(_ADJS), rng.choice(_NOUNS)]
    # vary length a little
    if rng.random() < 0.5:
        words = words[:5] + ['.']
    else:
        words = words + ['.']
    return ' '.join(words[:-1]) + '.'


# ---- learnable mode: the caption is a deterministic function of the ----
# image.  Slot choices drawn from RandomState(image_id) pick both the
# caption words and the image's horizontal color bands, so a model that
# reads the bands through the (random, frozen) CNN features can genuinely
# generalize to held-out image ids — the honest synthetic analog of the
# reference's COCO BLEU protocol.

_SLOTS = [_ADJS, _NOUNS, _VERBS, _PREPS, _ADJS, _NOUNS]


def _learnable_choices(image_id):
    rng = np.random.RandomState(image_id % (2 ** 31))
    return [int(rng.randint(len(pool))) for pool in _SLOTS]


def learnable_caption(image_id):
    c = _learnable_choices(image_id)
    return 'a %s %s %s %s a %s %s.' % (
        _SLOTS[0][c[0]], _SLOTS[1][c[1]], _SLOTS[2][c[2]],
        _SLOTS[3][c[3]], _SLOTS[4][c[4]], _SLOTS[5][c[5]])


def learnable_image(image_id, shape=(224, 224, 3)):
    """Horizontal bands encode the slot choices (values sized like
    mean-subtracted pixels)."""
    h, w, _ = shape
    img = np.zeros(shape, dtype=np.float32)
    band = h // len(_SLOTS)
    c = _learnable_choices(image_id)
    for i, (choice, pool) in enumerate(zip(c, _SLOTS)):
        frac = (choice + 1) / (len(pool) + 1)
        y0, y1 = i * band, (i + 1) * band if i + 1 < len(_SLOTS) else h
        img[y0:y1, :, 0] = 200.0 * frac - 100.0
        img[y0:y1, :, 1] = 200.0 * ((choice * 7) % len(pool)) \
            / len(pool) - 100.0
        img[y0:y1, :, 2] = 100.0 * np.sin(choice + i)
    return img


def make_coco_dict(num_images, caps_per_image, seed, prefix=0,
                   mode='noise'):
    rng = np.random.RandomState(seed)

    class _R:  # adapter: RandomState with python-like choice/random
        def choice(self, xs):
            return xs[rng.randint(len(xs))]

        def random(self):
            return rng.rand()

    r = _R()
    images, annotations = [], []
    ann_id = 1
    for k in range(num_images):
        iid = prefix * 10 ** 6 + k + 1
        images.append({'id': iid,
                       'file_name': 'synthetic://%d' % iid})
        for _ in range(caps_per_image):
            cap = (learnable_caption(iid) if mode == 'learnable'
                   else _make_caption(r))
            annotations.append({'id': ann_id, 'image_id': iid,
                                'caption': cap})
            ann_id += 1
    return {'images': images, 'annotations': annotations}


def make_coco(num_images, caps_per_image, seed, prefix=0, mode='noise'):
    coco = COCO()
    coco.dataset = make_coco_dict(num_images, caps_per_image, seed, prefix,
                                  mode)
    coco.process_dataset()
    coco.createIndex()
    return coco


class SyntheticImageLoader(object):
    """Deterministic 'images' for synthetic:// files.

    mode 'noise': id-seeded gaussian noise (pipeline tests).
    mode 'learnable': band-coded images matching learnable_caption(id)."""

    def __init__(self, image_shape=(224, 224, 3), seed=0, mode='noise',
                 cache_limit=2048):
        self.image_shape = tuple(image_shape)
        self.seed = seed
        self.mode = mode
        self._cache = {}
        self._cache_limit = cache_limit

    def load_image(self, image_file):
        iid = int(str(image_file).split('://')[-1])
        img = self._cache.get(iid)
        if img is not None:
            return img
        if self.mode == 'learnable':
            img = learnable_image(iid, self.image_shape)
        else:
            rng = np.random.RandomState(
                (self.seed * 1000003 + iid) % (2 ** 31))
            img = rng.randn(*self.image_shape).astype(np.float32) * 50.0
        if len(self._cache) < self._cache_limit:
            self._cache[iid] = img
        return img

    def load_images(self, image_files):
        return np.stack([self.load_image(f) for f in image_files], axis=0)


def _vocab_from_coco(config, coco):
    vocabulary = Vocabulary(config.vocabulary_size)
    vocabulary.build(coco.all_captions())
    config.vocabulary_size = vocabulary.size
    return vocabulary


def prepare_train_data(config):
    n = getattr(config, 'synthetic_num_images', 640)
    mode = getattr(config, 'synthetic_mode', 'noise')
    coco = make_coco(n, 1, config.seed, mode=mode)
    coco.filter_by_cap_len(config.max_caption_length)
    vocabulary = _vocab_from_coco(config, coco)
    coco.filter_by_words(set(vocabulary.words))

    ann_ids = list(coco.anns.keys())
    if config.max_train_ann_num:
        ann_ids = ann_ids[:config.max_train_ann_num]
    captions = [coco.anns[a]['caption'] for a in ann_ids]
    image_ids = [coco.anns[a]['image_id'] for a in ann_ids]
    image_files = [coco.imgs[i]['file_name'] for i in image_ids]
    word_idxs, masks = _tensorize_captions(
        captions, vocabulary, config.max_caption_length)
    return DataSet(image_ids, image_files, config.batch_size,
                   word_idxs, masks, True, True)


def prepare_eval_data(config):
    n = min(getattr(config, 'synthetic_num_images', 640), 64)
    mode = getattr(config, 'synthetic_mode', 'noise')
    coco = make_coco(n, 5, config.seed + 1, prefix=1, mode=mode)
    if config.max_eval_ann_num:
        ann_ids = list(coco.anns.keys())[:config.max_eval_ann_num]
        image_ids = list(dict.fromkeys(
            coco.anns[a]['image_id'] for a in ann_ids))
    else:
        image_ids = list(coco.imgs.keys())
    image_files = [coco.imgs[i]['file_name'] for i in image_ids]
    # build vocabulary from a train-shaped corpus so idx<->word matches train
    train_coco = make_coco(getattr(config, 'synthetic_num_images', 640), 1,
                           config.seed, mode=mode)
    vocabulary = _vocab_from_coco(config, train_coco)
    dataset = DataSet(image_ids, image_files, config.batch_size)
    return coco, dataset, vocabulary


def prepare_test_data(config):
    n = 8
    coco = make_coco(n, 1, config.seed + 2, prefix=2)
    image_ids = list(coco.imgs.keys())
    image_files = [coco.imgs[i]['file_name'] for i in image_ids]
    train_coco = make_coco(getattr(config, 'synthetic_num_images', 640), 1,
                           config.seed)
    vocabulary = _vocab_from_coco(config, train_coco)
    dataset = DataSet(image_ids, image_files, config.batch_size)
    return dataset, vocabulary
