from .eval import COCOEvalCap
