from .datasets import (
    ObjectAttributeDataset, SynthDataset, SyntheticImageDataset,
    collate_fn, get_classnames, insert_rand_word,
)
from .tokenizer import HashTokenizer, load_tokenizer
from .transforms import TrainTransform, EvalTransform
from .augment import rand_bbox, cutmix_data, mixup_data, mixup_criterion

__all__ = [
    "ObjectAttributeDataset", "SynthDataset", "SyntheticImageDataset",
    "collate_fn", "get_classnames", "insert_rand_word",
    "HashTokenizer", "load_tokenizer", "TrainTransform", "EvalTransform",
    "rand_bbox", "cutmix_data", "mixup_data", "mixup_criterion",
]
