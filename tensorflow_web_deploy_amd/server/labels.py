"""Class-id -> human label mapping (reference: digit/synset lookup table).

MNIST digits for the LeNet config; for ImageNet-class models with no
label file we synthesize "class_NNN" names (benchmarks use random-init
weights, so real synset names carry no information here). A user label
file (one label per line, index = class id) overrides both.
"""
from __future__ import annotations

from typing import List, Optional

MNIST_LABELS = [str(i) for i in range(10)]


def load_labels(path: Optional[str], num_classes: int) -> List[str]:
    if path:
        with open(path) as fh:
            labels = [ln.rstrip("\n") for ln in fh if ln.strip()]
        if len(labels) >= num_classes:
            return labels[:num_classes]
        labels += [f"class_{i}" for i in range(len(labels), num_classes)]
        return labels
    if num_classes == 10:
        return MNIST_LABELS
    return [f"class_{i}" for i in range(num_classes)]
