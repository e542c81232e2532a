import numpy as np

from sat_amd.data.dataset import DataSet
from sat_amd.data.synthetic import SyntheticImageLoader


def _ds(n=7, bs=3, train=True):
    ids = list(range(n))
    files = ['synthetic://%d' % (i + 1) for i in range(n)]
    wi = np.arange(n * 4).reshape(n, 4).astype(np.int32)
    mk = np.ones((n, 4), dtype=np.float32)
    return DataSet(ids, files, bs, wi, mk, is_train=train, shuffle=False)


def test_batch_counts():
    ds = _ds()
    assert ds.num_batches == 3
    assert ds.fake_count == 2


def test_iteration_and_padding():
    ds = _ds()
    seen = 0
    while ds.has_next_batch():
        files, wi, mk = ds.next_batch()
        assert len(files) == 3 and wi.shape == (3, 4)
        seen += 1
    assert seen == 3


def test_partial_batch_resamples_from_dataset():
    ds = _ds()
    ds.next_batch()
    ds.next_batch()
    files, wi, mk = ds.next_batch()  # 1 real + 2 fake
    assert all(f in ds.image_files for f in files)


def test_reset_restarts():
    ds = _ds()
    while ds.has_next_batch():
        ds.next_batch()
    assert not ds.has_next_batch()
    ds.reset()
    assert ds.has_next_batch()


def test_eval_mode_returns_files_only():
    ds = _ds(train=False)
    out = ds.next_batch()
    assert isinstance(out, np.ndarray) and out.shape == (3,)


def test_shuffle_changes_order():
    np.random.seed(0)
    ds = DataSet(list(range(50)), ['f%d' % i for i in range(50)], 10,
                 np.zeros((50, 4)), np.zeros((50, 4)), True, shuffle=True)
    assert ds.idxs != list(range(50))


def test_synthetic_loader_deterministic():
    ld = SyntheticImageLoader((8, 8, 3), seed=5)
    a = ld.load_image('synthetic://7')
    b = ld.load_image('synthetic://7')
    c = ld.load_image('synthetic://8')
    assert np.array_equal(a, b)
    assert not np.array_equal(a, c)
    batch = ld.load_images(['synthetic://7', 'synthetic://8'])
    assert batch.shape == (2, 8, 8, 3)
