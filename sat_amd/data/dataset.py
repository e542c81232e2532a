"""Dataset preparation and the in-memory batcher.

Behavioral parity with reference `dataset.py`:
  * `DataSet` — index-shuffling batcher over numpy arrays of (image_ids,
    image_files, word_idxs[N,T] int32, masks[N,T] float32); the last partial
    batch is padded with `fake_count` random resamples (dataset.py:52-54);
    train batches return (image_files, word_idxs, masks), eval/test batches
    return image_files only (dataset.py:56-64).
  * `prepare_train_data` (dataset.py:74-169): COCO load with annotation cap →
    caption-length filter → vocabulary build-or-load → vocabulary filter →
    tokenize+pad captions with masks → cache to anns.csv + data.npy →
    shuffled train DataSet.
  * `prepare_eval_data` (dataset.py:171-205), `prepare_test_data`
    (dataset.py:207-226), `build_vocabulary` (dataset.py:228-239).

Additions: `config.synthetic_data` routes every prepare_* through
`sat_amd.data.synthetic`, which fabricates a COCO-shaped corpus + images in
memory — no network, no files — so the full train/eval/test pipeline runs
hermetically (the reference instead downloads COCO images per run).
"""

import os

import numpy as np
import pandas as pd

from .coco import COCO
from .vocabulary import Vocabulary


class DataSet(object):
    def __init__(self, image_ids, image_files, batch_size, word_idxs=None,
                 masks=None, is_train=False, shuffle=False):
        self.image_ids = np.array(image_ids)
        self.image_files = np.array(image_files)
        self.word_idxs = np.array(word_idxs) if word_idxs is not None else None
        self.masks = np.array(masks) if masks is not None else None
        self.batch_size = batch_size
        self.is_train = is_train
        self.shuffle = shuffle
        self.setup()

    def setup(self):
        self.count = len(self.image_ids)
        self.num_batches = int(np.ceil(self.count / float(self.batch_size)))
        self.fake_count = self.num_batches * self.batch_size - self.count
        self.idxs = list(range(self.count))
        self.reset()

    def reset(self):
        self.current_idx = 0
        if self.shuffle:
            np.random.shuffle(self.idxs)

    def next_batch(self):
        assert self.has_next_batch()
        if self.has_full_next_batch():
            current_idxs = self.idxs[self.current_idx:
                                     self.current_idx + self.batch_size]
        else:
            current_idxs = (self.idxs[self.current_idx:self.count]
                            + list(np.random.choice(self.count,
                                                    self.fake_count)))
        image_files = self.image_files[current_idxs]
        self.current_idx += self.batch_size
        if self.is_train:
            return (image_files, self.word_idxs[current_idxs],
                    self.masks[current_idxs])
        return image_files

    def has_next_batch(self):
        return self.current_idx < self.count

    def has_full_next_batch(self):
        return self.current_idx + self.batch_size <= self.count


def _tensorize_captions(captions, vocabulary, max_len):
    """Tokenize + left-align + zero-pad captions; build float masks."""
    word_idxs = np.zeros((len(captions), max_len), dtype=np.int32)
    masks = np.zeros((len(captions), max_len), dtype=np.float32)
    for i, caption in enumerate(captions):
        idxs = vocabulary.process_sentence(caption)[:max_len]
        word_idxs[i, :len(idxs)] = idxs
        masks[i, :len(idxs)] = 1.0
    return word_idxs, masks


def prepare_train_data(config):
    """Prepare the data for training the model."""
    if getattr(config, 'synthetic_data', False):
        from . import synthetic
        return synthetic.prepare_train_data(config)

    coco = COCO(config.train_caption_file, config.max_train_ann_num)
    coco.filter_by_cap_len(config.max_caption_length)

    print("Building the vocabulary...")
    vocabulary = Vocabulary(config.vocabulary_size)
    if not os.path.exists(config.vocabulary_file):
        caps = coco.all_captions()
        if config.max_train_ann_num:
            caps = caps[:config.max_train_ann_num]
        vocabulary.build(caps)
        vocabulary.save(config.vocabulary_file)
    else:
        vocabulary.load(config.vocabulary_file)
    print("Vocabulary built.")
    print("Number of words = %d" % vocabulary.size)
    config.vocabulary_size = vocabulary.size

    coco.filter_by_words(set(vocabulary.words))

    print("Processing the captions...")
    if not os.path.exists(config.temp_annotation_file):
        ann_ids = list(coco.anns.keys())
        if config.max_train_ann_num:
            ann_ids = ann_ids[:config.max_train_ann_num]
        captions = [coco.anns[a]['caption'] for a in ann_ids]
        image_ids = [coco.anns[a]['image_id'] for a in ann_ids]
        image_files = [os.path.join(config.train_image_dir,
                                    coco.imgs[i]['file_name'])
                       for i in image_ids]
        pd.DataFrame({'image_id': image_ids, 'image_file': image_files,
                      'caption': captions}
                     ).to_csv(config.temp_annotation_file)
    else:
        annotations = pd.read_csv(config.temp_annotation_file)
        n = config.max_train_ann_num or len(annotations)
        captions = annotations['caption'].values[:n]
        image_ids = annotations['image_id'].values[:n]
        image_files = annotations['image_file'].values[:n]

    if not os.path.exists(config.temp_data_file):
        word_idxs, masks = _tensorize_captions(
            captions, vocabulary, config.max_caption_length)
        np.save(config.temp_data_file,
                {'word_idxs': word_idxs, 'masks': masks})
    else:
        data = np.load(config.temp_data_file, allow_pickle=True,
                       encoding='latin1').item()
        word_idxs, masks = data['word_idxs'], data['masks']
    print("Captions processed.")
    print("Number of captions = %d" % len(captions))

    # fetch any images the annotation set references but the disk lacks
    # (reference dataset.py:157 -> coco.download); offline this degrades
    # to a counted-failure no-op
    missing = [int(i) for i, f in zip(image_ids, image_files)
               if not os.path.exists(f)]
    if missing:
        coco.download(config.train_image_dir, list(set(missing)))

    dataset = DataSet(image_ids, image_files, config.batch_size,
                      word_idxs, masks, True, True)
    return dataset


def prepare_eval_data(config):
    """Prepare the data for evaluating the model."""
    if getattr(config, 'synthetic_data', False):
        from . import synthetic
        return synthetic.prepare_eval_data(config)

    coco = COCO(config.eval_caption_file, config.max_eval_ann_num)
    if config.max_eval_ann_num:
        ann_ids = list(coco.anns.keys())[:config.max_eval_ann_num]
        image_ids = list(dict.fromkeys(
            coco.anns[a]['image_id'] for a in ann_ids))
    else:
        image_ids = list(coco.imgs.keys())
    image_files = [os.path.join(config.eval_image_dir,
                                coco.imgs[i]['file_name'])
                   for i in image_ids]

    print("Building the vocabulary...")
    if os.path.exists(config.vocabulary_file):
        vocabulary = Vocabulary(config.vocabulary_size,
                                config.vocabulary_file)
    else:
        vocabulary = build_vocabulary(config)
    print("Number of words = %d" % vocabulary.size)
    config.vocabulary_size = vocabulary.size

    dataset = DataSet(image_ids, image_files, config.batch_size)
    return coco, dataset, vocabulary


def prepare_test_data(config):
    """Prepare the data for testing the model."""
    if getattr(config, 'synthetic_data', False):
        from . import synthetic
        return synthetic.prepare_test_data(config)

    files = os.listdir(config.test_image_dir)
    image_files = [os.path.join(config.test_image_dir, f) for f in files
                   if f.lower().endswith(('.jpg', '.jpeg'))]
    image_ids = list(range(len(image_files)))

    print("Building the vocabulary...")
    if os.path.exists(config.vocabulary_file):
        vocabulary = Vocabulary(config.vocabulary_size,
                                config.vocabulary_file)
    else:
        vocabulary = build_vocabulary(config)
    print("Number of words = %d" % vocabulary.size)
    config.vocabulary_size = vocabulary.size

    dataset = DataSet(image_ids, image_files, config.batch_size)
    return dataset, vocabulary


def build_vocabulary(config, max_ann_num=None):
    """Build the vocabulary from the training data and save it to a file."""
    coco = COCO(config.train_caption_file, config.max_train_ann_num)
    coco.filter_by_cap_len(config.max_caption_length)
    vocabulary = Vocabulary(config.vocabulary_size)
    caps = coco.all_captions()
    if config.max_train_ann_num:
        caps = caps[:config.max_train_ann_num]
    vocabulary.build(caps)
    vocabulary.save(config.vocabulary_file)
    return vocabulary
