"""CLI / config for cilfw.

Reproduces the reference's flag surface (reference ``template.py:13-49``,
``get_args_parser``) so existing run commands keep working, plus cilfw-native flags for
the MI355X build (dtype, backend selection, checkpoint dir, synthetic data).

Like the reference, the parsed Namespace doubles as the mutable run state: the
orchestrator attaches ``class_order``, ``nb_classes``, ``task_id``, ``known_classes``
and ``increment_per_task`` during the run (reference ``template.py:201-303``).
"""

import argparse


def get_args_parser():
    parser = argparse.ArgumentParser("cilfw class-incremental training", add_help=False)

    # reproducibility (reference default: seed=0, template.py:16)
    parser.add_argument("--seed", default=0, type=int)

    # CIL protocol (reference template.py:16-19)
    parser.add_argument("--num_bases", default=50, type=int,
                        help="classes in the base (first) task")
    parser.add_argument("--increment", default=10, type=int,
                        help="classes added per subsequent task")

    # model / input (reference template.py:20-22)
    parser.add_argument("--backbone", default="resnet32", type=str,
                        choices=["resnet20", "resnet32", "resnet44", "resnet56",
                                 "resnet110", "resnet18", "resnet34", "resnet50"])
    parser.add_argument("--batch_size", default=128, type=int, help="per-GPU batch size")
    parser.add_argument("--input_size", default=32, type=int)

    # augmentation (reference template.py:23-33)
    parser.add_argument("--color_jitter", type=float, default=0.4)
    parser.add_argument("--aa", type=str, default="rand-m9-mstd0.5-inc1",
                        help='RandAugment policy ("" disables)')
    parser.add_argument("--train_interpolation", type=str, default="bicubic")
    parser.add_argument("--reprob", type=float, default=0.0,
                        help="random erasing prob (reference default 0.0)")
    parser.add_argument("--remode", type=str, default="pixel")
    parser.add_argument("--recount", type=int, default=1)
    parser.add_argument("--resplit", action="store_true", default=False)

    # rehearsal memory (reference template.py:34-36)
    parser.add_argument("--herding_method", default="barycenter", type=str,
                        choices=["barycenter", "random"])
    parser.add_argument("--memory_size", default=2000, type=int)
    parser.add_argument("--fixed_memory", action="store_true", default=False)

    # optimization (reference template.py:37-42)
    parser.add_argument("--lr", default=0.1, type=float)
    parser.add_argument("--momentum", default=0.9, type=float)
    parser.add_argument("--weight_decay", default=5e-4, type=float)
    parser.add_argument("--num_epochs", default=140, type=int)
    parser.add_argument("--smooth", default=0.0, type=float, help="label smoothing")
    parser.add_argument("--eval_every_epoch", default=5, type=int)

    # distributed (reference template.py:43-44)
    parser.add_argument("--dist_url", default="env://", type=str)

    # dataset (reference template.py:45-46)
    parser.add_argument("--data_set", default="cifar100", type=str,
                        choices=["cifar100", "imagenet100", "imagenet1000", "cub200",
                                 "synthetic", "synthetic_hard"])
    parser.add_argument("--data_path", default="./data/cifar100", type=str)

    # distillation (reference template.py:47-48)
    parser.add_argument("--lambda_kd", default=0.5, type=float)
    parser.add_argument("--dynamic_lambda_kd", action="store_true", default=False,
                        help="scale lambda_kd by known/(known+new) per task "
                             "(the reference parsed but never used this flag — "
                             "template.py:48; here it is honored)")
    parser.add_argument("--kd_temperature", default=2.0, type=float)

    # ---- cilfw-native flags (no reference counterpart) ----
    parser.add_argument("--class_order", default=None, type=str,
                        help="comma-separated class permutation; default: the "
                             "reference's hardcoded CIFAR-100 order "
                             "(template.py:201-202) for cifar100, else a "
                             "seeded permutation")
    parser.add_argument("--dtype", default="bf16", type=str, choices=["bf16", "fp32"],
                        help="compute dtype for the backbone")
    parser.add_argument("--device", default=None, type=str,
                        help="cuda|cpu (default: cuda if available)")
    parser.add_argument("--workers", default=4, type=int)
    parser.add_argument("--output_dir", default="", type=str,
                        help="checkpoint directory ('' disables checkpointing)")
    parser.add_argument("--resume", default="", type=str,
                        help="resume from a per-task checkpoint file")
    parser.add_argument("--synthetic_classes", default=100, type=int,
                        help="class count for --data_set synthetic")
    parser.add_argument("--synthetic_train_size", default=5000, type=int,
                        help="samples per class split across classes for --data_set "
                             "synthetic")
    parser.add_argument("--no_aug", action="store_true", default=False)
    parser.add_argument("--ddp_bucket_mb", default=25.0, type=float,
                        help="gradient all-reduce bucket size (MB)")
    parser.add_argument("--gpu_data", action="store_true", default=False,
                        help="HBM-resident task data + on-device batch "
                             "assembly (crop/flip/normalize) — bypasses the "
                             "Python DataLoader for array-backed datasets")
    parser.add_argument("--no_wa", action="store_true", default=False,
                        help="ablation: skip the weight-align step")
    parser.add_argument("--no_replay", action="store_true", default=False,
                        help="ablation: no rehearsal exemplars (forgetting "
                             "baseline)")
    parser.add_argument("--no_device_replay", action="store_true", default=False,
                        help="with --gpu_data, source replay via host "
                             "add_samples instead of the HBM-resident "
                             "DeviceReplayMirror")
    parser.add_argument("--max_tasks", default=0, type=int,
                        help="stop after N tasks (0 = all) — e.g. the "
                             "BASELINE config[0] 2-task plumbing oracle")
    parser.add_argument("--metric_every", default=1, type=int,
                        help="read train metrics to host every N steps "
                             "(1 = reference-exact; higher avoids per-step "
                             "GPU syncs in the hot loop)")
    parser.add_argument("--no_step_graph", action="store_true", default=False,
                        help="disable hipGraph capture of the training step "
                             "(graphs are on by default with --gpu_data)")
    parser.add_argument("--dist_timeout", default=300.0, type=float,
                        help="collective timeout in seconds — a dead rank "
                             "raises instead of hanging (reference behavior "
                             "is an indefinite hang); also drives the "
                             "heartbeat watchdog")
    parser.add_argument("--compat_step_barrier", action="store_true", default=False,
                        help="reproduce the reference's per-training-step "
                             "dist.barrier (template.py:272) — a perf bug kept "
                             "behind a flag")
    return parser


def parse_args(argv=None):
    parser = argparse.ArgumentParser("cilfw", parents=[get_args_parser()])
    return parser.parse_args(argv)
