import json

from sat_amd.data.coco import COCO
from sat_amd.data.synthetic import make_coco, make_coco_dict


def _coco(tmp_path, **kw):
    d = make_coco_dict(4, 2, seed=0)
    f = tmp_path / 'caps.json'
    f.write_text(json.dumps(d))
    return COCO(str(f), **kw)


def test_index_built(tmp_path):
    c = _coco(tmp_path)
    assert len(c.imgs) == 4
    assert len(c.anns) == 8
    assert all(len(v) == 2 for v in c.imgToAnns.values())


def test_captions_normalized(tmp_path):
    c = _coco(tmp_path)
    for a in c.anns.values():
        assert a['caption'].endswith('.')
        assert a['caption'] == a['caption'].lower()


def test_max_ann_num_cap(tmp_path):
    c = _coco(tmp_path, max_ann_num=3)
    assert len(c.anns) == 3


def test_filter_by_cap_len():
    c = make_coco(6, 1, seed=1)
    before = len(c.anns)
    c.filter_by_cap_len(6)  # captions are 5 or 8 words + '.'
    assert 0 < len(c.anns) < before or len(c.anns) == before


def test_filter_by_words():
    c = make_coco(6, 1, seed=1)
    words = set()
    for a in list(c.anns.values())[:2]:
        from sat_amd.data.tokenizer import word_tokenize
        words.update(word_tokenize(a['caption']))
    c.filter_by_words(words)
    assert len(c.anns) >= 2


def test_load_res(tmp_path):
    c = _coco(tmp_path)
    results = [{'image_id': i, 'caption': 'a dog.'}
               for i in list(c.imgs.keys())[:2]]
    rf = tmp_path / 'res.json'
    rf.write_text(json.dumps(results))
    res = c.loadRes(str(rf))
    assert len(res.imgToAnns) == 2


def test_download_fetches_missing_images(tmp_path):
    """download() (reference coco.py:292-314 parity): fetches by
    coco_url, skips files already present, survives failures.  Tested
    offline via file:// URLs."""
    import json
    from sat_amd.data.coco import COCO

    src = tmp_path / 'src'
    src.mkdir()
    (src / 'img1.jpg').write_bytes(b'JPG1')
    (src / 'img2.jpg').write_bytes(b'JPG2')

    ann = {
        'images': [
            {'id': 1, 'file_name': 'img1.jpg',
             'coco_url': 'file://%s' % (src / 'img1.jpg')},
            {'id': 2, 'file_name': 'img2.jpg',
             'coco_url': 'file://%s' % (src / 'img2.jpg')},
            {'id': 3, 'file_name': 'img3.jpg',
             'coco_url': 'file://%s/does_not_exist.jpg' % src},
        ],
        'annotations': [
            {'id': 10, 'image_id': 1, 'caption': 'a cat .'},
            {'id': 11, 'image_id': 2, 'caption': 'a dog .'},
            {'id': 12, 'image_id': 3, 'caption': 'a bird .'},
        ],
        'type': 'captions',
    }
    af = tmp_path / 'anns.json'
    af.write_text(json.dumps(ann))

    tgt = tmp_path / 'imgs'
    tgt.mkdir()
    (tgt / 'img2.jpg').write_bytes(b'ALREADY_HERE')

    coco = COCO(str(af))
    done, failed = coco.download(str(tgt), retries=0)
    assert done == 1 and failed == 1
    assert (tgt / 'img1.jpg').read_bytes() == b'JPG1'
    # pre-existing file untouched (reference skips existing)
    assert (tgt / 'img2.jpg').read_bytes() == b'ALREADY_HERE'
    assert not (tgt / 'img3.jpg').exists()
