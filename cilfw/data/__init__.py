import numpy as np

from .scenario import ClassIncremental, TaskSet
from .datasets import build_source, make_synthetic, DATASET_STATS
from .transforms import TrainTransform, EvalTransform
from .sampler import DistributedSampler

# Fixed CIFAR-100 class-order permutation, as the reference hardcodes one
# (template.py:201-202) so runs are comparable across seeds.
CIFAR100_CLASS_ORDER = [
    87, 0, 52, 58, 44, 91, 68, 97, 51, 15, 94, 92, 10, 72, 49, 78, 61, 14, 8, 86,
    84, 96, 18, 24, 32, 45, 88, 11, 4, 67, 69, 66, 77, 47, 79, 93, 29, 50, 57, 83,
    17, 81, 41, 12, 37, 59, 25, 20, 80, 73, 1, 28, 6, 46, 62, 82, 53, 9, 31, 75,
    38, 63, 33, 74, 27, 22, 36, 3, 16, 21, 60, 19, 70, 90, 89, 43, 5, 42, 65, 76,
    40, 30, 23, 85, 2, 95, 56, 48, 71, 64, 98, 13, 99, 7, 34, 55, 54, 26, 35, 39,
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
    if class_order is None:
        if args.data_set.lower() == "cifar100" and nb_classes == 100:
            class_order = CIFAR100_CLASS_ORDER
        else:
            rng = np.random.default_rng(args.seed)
            class_order = rng.permutation(nb_classes).tolist()
        args.class_order = class_order
    scenario = ClassIncremental(x, y, args.num_bases, args.increment,
                                class_order=class_order, transform=transform)
    return scenario, nb_classes


__all__ = ["ClassIncremental", "TaskSet", "build_source", "make_synthetic",
           "build_dataset", "TrainTransform", "EvalTransform",
           "DistributedSampler", "DATASET_STATS", "CIFAR100_CLASS_ORDER"]
