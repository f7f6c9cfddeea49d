"""Process-group bootstrap — torchrun env contract, RCCL on GPU / gloo on CPU.

Reference utils.py:135-168: reads RANK/WORLD_SIZE/LOCAL_RANK, binds the device,
init_process_group('nccl', env://), barrier, rank-0 print gating. The reference
HARD-FAILS without torchrun env (utils.py:140-144); cilfw adds a single-process
fallback (world_size=1, no process group) so tests and bench run standalone.

On ROCm the "nccl" backend IS RCCL; collectives run over xGMI links.
"""

import builtins
import datetime
import os

import torch
import torch.distributed as dist


def setup_for_distributed(is_master):
    """Gate print() to rank 0 (reference utils.py:160-168)."""
    builtin_print = builtins.print

    def print_(*args, **kwargs):
        force = kwargs.pop("force", False)
        if is_master or force:
            builtin_print(*args, **kwargs)

    builtins.print = print_


def init_distributed_mode(args):
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        args.rank = int(os.environ["RANK"])
        args.world_size = int(os.environ["WORLD_SIZE"])
        args.local_rank = int(os.environ.get("LOCAL_RANK", args.rank))
    else:
        args.rank, args.world_size, args.local_rank = 0, 1, 0
        args.distributed = False
        if getattr(args, "device", None) is None:
            args.device = "cuda" if torch.cuda.is_available() else "cpu"
        return

    args.distributed = True
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(args.local_rank)  # bind BEFORE init (local, not
        # global rank — the reference's device_ids=[args.rank] was a
        # single-node-only bug, template.py:244)
        backend = "nccl"  # = RCCL on ROCm
        args.device = f"cuda:{args.local_rank}"
    else:
        backend = "gloo"
        if getattr(args, "device", None) is None:
            args.device = "cpu"
    # finite collective timeout — a dead peer surfaces as an error instead
    # of hanging forever (the reference hangs, template.py:272). For RCCL
    # the NCCL watchdog needs async error handling to turn a stuck
    # collective into an exception.
    timeout_s = float(getattr(args, "dist_timeout", 300.0) or 300.0)
    os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
    dist.init_process_group(backend=backend,
                            init_method=getattr(args, "dist_url", "env://"),
                            world_size=args.world_size, rank=args.rank,
                            timeout=datetime.timedelta(seconds=timeout_s))
    dist.barrier()
    setup_for_distributed(args.rank == 0)


def get_rank():
    return dist.get_rank() if dist.is_available() and dist.is_initialized() else 0


def get_world_size():
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() \
        else 1


def is_main_process():
    return get_rank() == 0


def barrier():
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
