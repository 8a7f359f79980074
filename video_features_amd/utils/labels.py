"""Label maps + the ``--show_pred`` top-5 printer.

Equivalent of the reference's ``show_predictions_on_dataset``
(reference utils/utils.py:19-47) over the public Kinetics-400 / ImageNet-1k
class lists (stored as JSON data files in this package).
"""
from __future__ import annotations

import functools
import json
import os
from typing import List

import torch

_HERE = os.path.dirname(__file__)


@functools.lru_cache(maxsize=None)
def class_names(dataset: str) -> List[str]:
    fname = {'kinetics': 'kinetics400.json', 'imagenet': 'imagenet1k.json'}[dataset]
    with open(os.path.join(_HERE, fname)) as f:
        return json.load(f)['classes']


def show_predictions_on_dataset(logits: torch.Tensor, dataset: str,
                                k: int = 5) -> List[str]:
    """Print (and return) the top-k classes with softmax scores for each row
    of ``logits`` ((B, C) or (C,))."""
    names = class_names(dataset)
    if logits.dim() == 1:
        logits = logits[None]
    probs = logits.float().softmax(dim=-1).mean(dim=0)
    topv, topi = probs.topk(k)
    lines = [f'{v.item():.5f} {names[i]}' for v, i in zip(topv, topi)]
    for ln in lines:
        print(ln)
    print()
    return lines
