from .beam import gen_sample
from .rouge import rouge_n, rouge_l, score_files

__all__ = ["gen_sample", "rouge_n", "rouge_l", "score_files"]
