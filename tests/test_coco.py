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
