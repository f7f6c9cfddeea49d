import numpy as np

from .scenario import ClassIncremental, TaskSet
from .datasets import (build_source, make_synthetic,
                       make_synthetic_hard, DATASET_STATS)
from .transforms import TrainTransform, EvalTransform
from .sampler import DistributedSampler

# The reference's exact hardcoded CIFAR-100 class-order permutation
# (template.py:201-202), copied verbatim so per-task accuracy trajectories are
# directly comparable against reference runs (it is a data constant, not code).
CIFAR100_CLASS_ORDER = [
    68, 56, 78, 8, 23, 84, 90, 65, 74, 76, 40, 89, 3, 92, 55, 9, 26, 80, 43, 38,
    58, 70, 77, 1, 85, 19, 17, 50, 28, 53, 13, 81, 45, 82, 6, 59, 83, 16, 15, 44,
    91, 41, 72, 60, 79, 52, 20, 10, 31, 54, 37, 95, 14, 71, 96, 98, 97, 2, 64, 66,
    42, 22, 35, 86, 24, 34, 87, 21, 99, 0, 88, 27, 18, 94, 11, 12, 47, 25, 30, 46,
    62, 69, 36, 61, 7, 63, 75, 5, 32, 4, 51, 48, 73, 93, 39, 67, 29, 49, 57, 33,
]


def build_dataset(is_train, args, transform="auto"):
    """(scenario, nb_classes) — the reference's build_dataset contract
    (utils.py:188-207, called template.py:203-204)."""
    x, y, nb_classes, stats_key = build_source(args, is_train)
    args._stats_key = stats_key
    if transform == "auto":
        transform = (TrainTransform(args, stats_key) if is_train
                     else EvalTransform(args, stats_key))
    class_order = getattr(args, "class_order", None)
    if isinstance(class_order, str):
        class_order = ([int(c) for c in class_order.split(",")]
                       if class_order else None)
    if class_order is None:
        if args.data_set.lower() == "cifar100" and nb_classes == 100:
            class_order = CIFAR100_CLASS_ORDER
        else:
            rng = np.random.default_rng(args.seed)
            class_order = rng.permutation(nb_classes).tolist()
    assert sorted(class_order) == list(range(nb_classes)), \
        "class_order must be a permutation of range(nb_classes)"
    args.class_order = class_order
    scenario = ClassIncremental(x, y, args.num_bases, args.increment,
                                class_order=class_order, transform=transform)
    return scenario, nb_classes


__all__ = ["ClassIncremental", "TaskSet", "build_source", "make_synthetic",
           "make_synthetic_hard",
           "build_dataset", "TrainTransform", "EvalTransform",
           "DistributedSampler", "DATASET_STATS", "CIFAR100_CLASS_ORDER"]
