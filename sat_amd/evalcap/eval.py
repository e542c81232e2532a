"""Caption-metric driver — interface parity with reference
`utils/coco/pycocoevalcap/eval.py:20-63` (COCOEvalCap.evaluate).

Computes Bleu_1..4, METEOR, ROUGE_L, CIDEr over tokenized ground-truth and
result captions; stores per-metric corpus scores in `.eval` and per-image
scores in `.imgToEval`, printing each corpus score as the reference does.
All scorers are pure Python (no Java subprocesses).
"""

from ..data.tokenizer import PTBTokenizer
from .bleu import Bleu
from .cider import Cider
from .meteor import Meteor
from .rouge import Rouge


class COCOEvalCap(object):
    def __init__(self, coco, cocoRes, eval_data=None):
        self.coco = coco
        self.cocoRes = cocoRes
        self.params = {'image_id': list(cocoRes.imgToAnns.keys())}
        self.eval = {}
        self.imgToEval = {}
        self.evalImgs = []

    def evaluate(self):
        img_ids = self.params['image_id']
        gts = {i: self.coco.imgToAnns[i] for i in img_ids}
        res = {i: self.cocoRes.imgToAnns[i] for i in img_ids}

        tokenizer = PTBTokenizer()
        gts = tokenizer.tokenize(gts)
        res = tokenizer.tokenize(res)

        scorers = [
            (Bleu(4), ["Bleu_1", "Bleu_2", "Bleu_3", "Bleu_4"]),
            (Meteor(), "METEOR"),
            (Rouge(), "ROUGE_L"),
            (Cider(), "CIDEr"),
        ]
        for scorer, method in scorers:
            score, scores = scorer.compute_score(gts, res)
            if isinstance(method, list):
                for sc, scs, m in zip(score, scores, method):
                    self.setEval(sc, m)
                    self.setImgToEvalImgs(scs, img_ids, m)
                    print("%s: %0.3f" % (m, sc))
            else:
                self.setEval(score, method)
                self.setImgToEvalImgs(scores, img_ids, method)
                label = method
                if method == "METEOR":
                    # exact+stem beam alignment only — no synonym /
                    # paraphrase stages (missing data files); NOT
                    # comparable to published METEOR numbers
                    label = "METEOR (approx: exact+stem stages)"
                print("%s: %0.3f" % (label, score))
        self.setEvalImgs()

    def setEval(self, score, method):
        self.eval[method] = score

    def setImgToEvalImgs(self, scores, img_ids, method):
        for iid, score in zip(img_ids, scores):
            self.imgToEval.setdefault(iid, {'image_id': iid})[method] = score

    def setEvalImgs(self):
        self.evalImgs = [self.imgToEval[i] for i in
                         sorted(self.imgToEval.keys())]
