"""CIL orchestrator — the per-task lifecycle (reference template.py:191-303).

Task loop: grow classifier -> wrap in the cilfw DP engine -> train (CE + KD vs the
frozen previous-task teacher) -> weight-align the new head -> cumulative eval ->
snapshot teacher -> herding feature pass -> update rehearsal memory -> checkpoint.

Printed schema matches the reference so runs are comparable:
per-epoch train meters, "* Acc@1 ... loss ..." eval lines, per-task
"task id = ...  @Acc1 = ..., acc1s = [...]".
"""

import random
import time

import numpy as np
import torch
from torch.utils.data import DataLoader

from . import ops
from .data import build_dataset, DistributedSampler, DATASET_STATS
from .data.gpu_pipeline import GpuTaskLoader
from .distributed import (init_distributed_mode, DataParallelEngine, barrier,
                          get_world_size, get_rank)
from .models import CilModel
from .cil import RehearsalMemory, save_task_checkpoint, load_task_checkpoint
from .optim import FlatSGD, CosineLR
from .utils.metrics import MetricLogger, SmoothedValue
from .utils.trace import trace_range


def init_seed(args):
    seed = args.seed
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def compute_dtype(args):
    if args.dtype == "bf16" and str(args.device).startswith("cuda"):
        return torch.bfloat16
    return torch.float32


def _gpu_data_active(args, device, dataset):
    """--gpu_data engages for array-backed datasets on cuda devices; the env
    override CILFW_GPU_DATA_ON_CPU=1 lets tests exercise the same loader /
    device-replay wiring on CPU (GpuTaskLoader is pure torch index ops)."""
    import os
    if not args.gpu_data or dataset.x.dtype != np.uint8:
        return False
    return (str(device).startswith("cuda")
            or os.environ.get("CILFW_GPU_DATA_ON_CPU") == "1")


def _to_device(inputs, targets, device, dtype):
    inputs = inputs.to(device, non_blocking=True).to(dtype)
    targets = targets.to(device, non_blocking=True)
    return inputs, targets


@torch.no_grad()
def evaluate(model, loader, device, args, header="Test:"):
    """Cumulative eval (reference template.py:169-188). Correct counts and the
    loss sum accumulate ON DEVICE (exact, not sampled) — one host sync at the
    end instead of per batch."""
    model.eval()
    dtype = compute_dtype(args)
    nb_classes = model.fc.nb_classes
    kmax = min(5, nb_classes)
    on_dev = str(device).startswith("cuda")
    dev = torch.device(device)
    correct = torch.zeros(2, dtype=torch.float64, device=dev)
    loss_sum = torch.zeros((), dtype=torch.float64, device=dev)
    n_total = 0
    n_batches = 0
    for inputs, targets, _tids in loader:
        inputs, targets = _to_device(inputs, targets, device, dtype)
        logits, _ = model(inputs)
        loss, _ce, _kd = ops.wa_loss(logits, None, targets)
        bs = targets.shape[0]
        if on_dev:
            from .ops._backend import use_hip, ext
            if use_hip(logits):
                counts = ext().topk_correct(logits, targets, kmax)
                correct[0] += counts[0]
                correct[1] += counts[kmax - 1]
            else:
                a = ops.accuracy(logits, targets, topk=(1, kmax))
                correct[0] += a[0] * bs / 100.0
                correct[1] += a[1] * bs / 100.0
        else:
            a = ops.accuracy(logits, targets, topk=(1, kmax))
            correct[0] += a[0] * bs / 100.0
            correct[1] += a[1] * bs / 100.0
        loss_sum += loss.double()
        n_total += bs
        n_batches += 1
    # exact cross-rank sample-weighted means (reference utils.py:36-43)
    metric_logger = MetricLogger()
    metric_logger.update_n(n=n_total, acc1=correct[0].item() * 100.0
                           / max(n_total, 1))
    metric_logger.update_n(n=n_total, acc5=correct[1].item() * 100.0
                           / max(n_total, 1))
    metric_logger.update_n(n=n_batches, loss=loss_sum.item()
                           / max(n_batches, 1))
    metric_logger.synchronize_between_processes(device=dev)
    acc1 = metric_logger.meters["acc1"].global_avg
    acc5 = metric_logger.meters["acc5"].global_avg
    lossv = metric_logger.meters["loss"].global_avg
    print(f"* Acc@1 {acc1:.3f} Acc@5 {acc5:.3f} loss {lossv:.3f}")
    model.train()
    return acc1


class _GraphedStep:
    """hipGraph-captured training step for the in-engine hot loop: the whole
    student fwd + teacher fwd + CE/KD + backward + fused SGD replays as one
    graph launch. Batches are copied into static input buffers; outputs
    (losses, logits) are static storages read after each replay. Re-captured
    per epoch (the cosine LR is baked into the captured SGD launch)."""

    def __init__(self, step_fn, x0, y0, model):
        # torch.cuda.graph REQUIRES a side-stream warmup before capture. The
        # warmup runs fwd+bwd WITHOUT the optimizer step and with BN running
        # stats snapshot/restored, so it leaves no trace on training state.
        # Stream capture RECORDS kernels without executing them, so after
        # capture we replay the graph once on (x0, y0) — that replay IS the
        # capture batch's training step and fills self.out with real values.
        self.static_x = x0.clone()
        self.static_y = y0.clone()
        stats = [(b, b.clone()) for n, b in model.named_buffers()
                 if "running" in n]
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            step_fn(self.static_x, self.static_y, update=False)
        torch.cuda.current_stream().wait_stream(s)
        for b, saved in stats:
            b.copy_(saved)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = step_fn(self.static_x, self.static_y)
        self.graph.replay()  # execute the recorded step for the capture batch

    def __call__(self, x, y):
        self.static_x.copy_(x)
        self.static_y.copy_(y)
        self.graph.replay()
        return self.out


def train_one_task(model, teacher, engine, optimizer, scheduler, train_loader,
                   train_sampler, val_loader, device, args):
    dtype = compute_dtype(args)
    known = args.known_classes
    lambda_kd = args.lambda_kd
    if args.dynamic_lambda_kd and known > 0:
        # n/(n+m) scaling the reference documented but never wired
        # (README.md:175-176 / dead flag template.py:48)
        lambda_kd = known / (known + args.increment_per_task)
    # frozen-teacher forward overlaps the student forward on a side HIP stream
    # (independent until the KD loss joins them; capture-compatible fork/join)
    tstream = (torch.cuda.Stream()
               if (teacher is not None and str(device).startswith("cuda"))
               else None)

    def step_fn(inputs, targets, update=True):
        optimizer.zero_grad()
        if tstream is not None:
            tstream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(tstream), torch.no_grad():
                t_logits, _ = teacher(inputs)
        logits, _features = model(inputs)
        if teacher is not None:
            if tstream is not None:
                torch.cuda.current_stream().wait_stream(tstream)
            else:
                with torch.no_grad():
                    t_logits, _ = teacher(inputs)
            loss, loss_ce, loss_kd = ops.wa_loss(
                logits, t_logits, targets, args.smooth,
                args.kd_temperature, lambda_kd)
        else:
            loss, loss_ce, loss_kd = ops.wa_loss(
                logits, None, targets, args.smooth,
                args.kd_temperature, lambda_kd)
        loss.backward()
        engine.finalize()
        if update:
            optimizer.step()
        return logits, loss_ce, loss_kd, loss

    import os as _os
    can_graph = (str(device).startswith("cuda") and not args.no_step_graph
                 and not args.compat_step_barrier
                 and not isinstance(train_loader, DataLoader)
                 and (get_world_size() == 1
                      or _os.environ.get("CILFW_GRAPH_MULTI") == "1"))
    model.train()
    graphed = None  # ONE capture per task: the lr is a device-tensor input
    # to the captured SGD launch (FlatSGD._lr_dev), so the cosine schedule
    # updates it by a 4-byte fill instead of forcing per-epoch re-capture
    for epoch in range(args.num_epochs):
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        elif hasattr(train_loader, "set_epoch"):
            train_loader.set_epoch(epoch)
        epoch_range = trace_range(f"task{args.task_id}/epoch{epoch}")
        epoch_range.__enter__()
        metric_logger = MetricLogger()
        metric_logger.meters["lr"] = SmoothedValue(fmt="{value:.6f}")
        t0 = time.time()
        nimg = 0
        first_of_epoch = epoch == 0
        metric_every = max(int(getattr(args, "metric_every", 1)), 1)
        step_i = 0
        wd = getattr(args, "_watchdog", None)
        for inputs, targets, _tids in train_loader:
            if wd is not None:
                wd.beat()
            inputs, targets = _to_device(inputs, targets, device, dtype)
            if can_graph and graphed is None and not first_of_epoch:
                # capture on this batch; the capture run IS its training step.
                # Drop the previous batch's autograd graph first: live loss/
                # logits refs keep AccumulateGrad nodes pinned to the default
                # stream, which breaks (segfaults) stream capture.
                logits = loss_ce = loss_kd = loss = None  # noqa: F841
                try:
                    graphed = _GraphedStep(step_fn, inputs, targets, model)
                    logits, loss_ce, loss_kd, loss = graphed.out
                except Exception as e:
                    print(f"[engine] step-graph capture failed "
                          f"({type(e).__name__}: {e}); running eager")
                    can_graph = False
                    logits, loss_ce, loss_kd, loss = step_fn(inputs, targets)
            elif graphed is not None:
                logits, loss_ce, loss_kd, loss = graphed(inputs, targets)
            else:
                logits, loss_ce, loss_kd, loss = step_fn(inputs, targets)
                if args.compat_step_barrier:
                    barrier()  # reference per-step barrier (template.py:272)
            first_of_epoch = False
            bs = targets.shape[0]
            nimg += bs
            if step_i % metric_every == 0:  # host sync point (reads scalars)
                accs = ops.accuracy(logits, targets,
                                    topk=(1, min(5, logits.shape[1])))
                metric_logger.update(ce=loss_ce.item(), kd=loss_kd.item(),
                                     loss=loss.item())
                metric_logger.update_n(n=bs, acc1=accs[0])
                metric_logger.meters["lr"].update(optimizer.lr)
            step_i += 1
        epoch_range.__exit__(None, None, None)
        metric_logger.synchronize_between_processes(device=torch.device(device))
        scheduler.step()
        ips = nimg * get_world_size() / max(time.time() - t0, 1e-9)
        print(f"task {args.task_id} epoch {epoch}: {metric_logger}  "
              f"imgs/s {ips:.0f}")
        if args.eval_every_epoch and (epoch + 1) % int(args.eval_every_epoch) == 0 \
                and epoch + 1 < args.num_epochs:
            evaluate(model, val_loader, device, args)


@torch.no_grad()
def extract_task_features(model, dataset, device, args):
    """Unshuffled, NON-distributed feature pass over the full task set — the
    reference runs this replicated on every rank (template.py:292-299); herding is
    deterministic so the resulting memory is rank-identical."""
    model.eval()
    dtype = compute_dtype(args)
    if _gpu_data_active(args, device, dataset):
        mean, std = DATASET_STATS[getattr(args, "_stats_key", "synthetic")]
        loader = GpuTaskLoader(dataset, args.batch_size, device, mean, std,
                               shuffle=False, augment=False, drop_last=False,
                               dtype=dtype)
    else:
        # un-augmented features on the CPU path too (the device path always
        # extracts with augment=False; advisor round-1 flagged the mismatch)
        from .data.scenario import TaskSet
        from .data.transforms import EvalTransform
        ds_eval = TaskSet(dataset.x, dataset.y, dataset.t,
                          EvalTransform(args, getattr(args, "_stats_key",
                                                      "synthetic")))
        loader = DataLoader(ds_eval, batch_size=args.batch_size,
                            shuffle=False, num_workers=args.workers,
                            drop_last=False)
    feats = []
    for inputs, _targets, _tids in loader:
        inputs = inputs.to(device, non_blocking=True).to(dtype)
        feats.append(model.extract_vector(inputs).float())
    model.train()
    return torch.cat(feats)


def run(args):
    init_distributed_mode(args)
    init_seed(args)
    device = args.device

    # failure detection (reference has none — a dead rank hangs the job):
    # collectives time out via the process-group timeout; the heartbeat
    # watchdog covers non-collective hangs. Watchdog fires a bit after the
    # collective timeout so the cleaner error path wins when both apply.
    from .distributed.watchdog import Watchdog, describe_failure
    watchdog = None
    if get_world_size() > 1:
        watchdog = Watchdog(
            float(getattr(args, "dist_timeout", 300.0) or 300.0) * 1.5,
            rank=get_rank(), checkpoint_dir=args.output_dir).start()
    args._watchdog = watchdog
    try:
        return _run_tasks(args, device, watchdog)
    except RuntimeError as e:
        if watchdog is not None:
            watchdog.stop()
        if get_world_size() > 1:
            import sys
            print(describe_failure(
                e, watchdog.last_checkpoint if watchdog else None),
                file=sys.stderr, flush=True)
            raise SystemExit(3)
        raise
    finally:
        if watchdog is not None:
            watchdog.stop()


def _run_tasks(args, device, watchdog):

    scenario_train, nb_classes = build_dataset(is_train=True, args=args)
    scenario_val, _ = build_dataset(is_train=False, args=args)
    args.nb_classes = nb_classes

    model = CilModel(args.backbone, args.input_size).to(device)
    barrier()

    memory = RehearsalMemory(args.memory_size, args.herding_method,
                             args.fixed_memory, nb_total_classes=nb_classes)
    teacher = None
    replay_mirror = None  # HBM-resident replay (DeviceReplayMirror)
    acc1s = []
    args.known_classes = 0
    start_task = 0

    if args.resume:
        state = load_task_checkpoint(args.resume, model, memory, args)
        model = model.to(device)
        acc1s = state["acc1s"]
        start_task = state["task_id"] + 1
        if start_task < len(scenario_train):
            teacher = model.copy().to(device)
            teacher.freeze(["all"])
            if compute_dtype(args) == torch.bfloat16:
                teacher.cast_compute_weights_(torch.bfloat16)
        print(f"resumed from {args.resume}: start_task={start_task}, "
              f"known={args.known_classes}, acc1s={acc1s}")

    nb_tasks = len(scenario_train)
    if getattr(args, "max_tasks", 0):
        nb_tasks = min(nb_tasks, args.max_tasks)
    for task_id in range(start_task, nb_tasks):
        args.task_id = task_id
        dataset_train = scenario_train[task_id]
        dataset_val = scenario_val[:task_id + 1]
        args.increment_per_task = scenario_train.increments(task_id)

        world, rank = get_world_size(), get_rank()
        use_gpu_data = _gpu_data_active(args, device, dataset_train)
        use_device_replay = (use_gpu_data
                             and not getattr(args, "no_device_replay", False))
        extra = None
        if task_id > 0 and getattr(args, "no_replay", False):
            pass  # ablation: train each task without rehearsal
        elif task_id > 0:
            if use_device_replay:
                # replay straight from the HBM-resident mirror — exemplars
                # never round-trip through the host (replaces the reference's
                # memory.get() -> numpy concat, template.py:230-231). The
                # mirror appends after the task samples exactly like
                # add_samples, so the shuffled index space is identical.
                if replay_mirror is None:  # resumed run: one rebuild upload
                    from .cil.replay_gpu import DeviceReplayMirror
                    replay_mirror = DeviceReplayMirror.from_memory(
                        memory, device)
                assert len(replay_mirror) == len(memory), \
                    "device replay mirror out of sync with rehearsal memory"
                extra = replay_mirror.get()
            else:
                mx, my, mt = memory.get()
                dataset_train.add_samples(mx, my, mt)

        if use_gpu_data:
            mean, std = DATASET_STATS[getattr(args, "_stats_key", "synthetic")]
            train_sampler = None
            # full device-side recipe: RandAugment/jitter pre-normalize +
            # random erasing post-normalize (cilfw/data/device_augment.py);
            # honors the same --aa/--color_jitter/--reprob flags as the host
            # pipeline (the round-1 device path silently dropped them)
            aug = None
            if not args.no_aug and (getattr(args, "aa", "")
                                    or getattr(args, "color_jitter", 0)
                                    or getattr(args, "reprob", 0)):
                from .data.device_augment import DeviceAugment
                aug = DeviceAugment(
                    aa_policy=getattr(args, "aa", ""),
                    color_jitter=getattr(args, "color_jitter", 0.0),
                    reprob=getattr(args, "reprob", 0.0),
                    remode=getattr(args, "remode", "pixel"),
                    recount=getattr(args, "recount", 1))
            train_loader = GpuTaskLoader(
                dataset_train, args.batch_size, device, mean, std,
                world=world, rank=rank, shuffle=True, seed=args.seed,
                augment=not args.no_aug, drop_last=True,
                dtype=compute_dtype(args), extra=extra, aug_pipeline=aug)
            val_loader = GpuTaskLoader(
                dataset_val, args.batch_size, device, mean, std,
                world=world, rank=rank, shuffle=False, augment=False,
                drop_last=False, dtype=compute_dtype(args))
        else:
            train_sampler = DistributedSampler(dataset_train, world, rank,
                                               shuffle=True, seed=args.seed)
            val_sampler = DistributedSampler(dataset_val, world, rank,
                                             shuffle=False)
            # reference-exact: the CPU path trains on the final partial batch
            # each epoch (template.py:236-239 has no drop_last); only the
            # graph-captured GpuTaskLoader path needs fixed batch shapes.
            train_loader = DataLoader(
                dataset_train, batch_size=args.batch_size,
                sampler=train_sampler, num_workers=args.workers,
                drop_last=False, persistent_workers=args.workers > 0)
            val_loader = DataLoader(dataset_val, batch_size=args.batch_size,
                                    sampler=val_sampler,
                                    num_workers=args.workers)

        model.prev_model_adaption(args.increment_per_task)
        engine = DataParallelEngine(model, bucket_mb=args.ddp_bucket_mb)
        optimizer = FlatSGD(engine, args.lr, args.momentum, args.weight_decay)
        scheduler = CosineLR(optimizer, t_max=args.num_epochs)

        train_one_task(model, teacher, engine, optimizer, scheduler,
                       train_loader, train_sampler, val_loader, device, args)

        if not getattr(args, "no_wa", False):
            model.after_model_adaption(args.increment_per_task, args)
        acc1 = evaluate(model, val_loader, device, args)
        acc1s.append(acc1)
        print(f"task id = {task_id}  @Acc1 = {acc1:.5f}, acc1s = {acc1s}")

        teacher = model.copy()
        teacher.freeze(["all"])
        if compute_dtype(args) == torch.bfloat16:
            teacher.cast_compute_weights_(torch.bfloat16)

        features = extract_task_features(model, dataset_train, device, args)
        rx, ry, rt = dataset_train.get_raw_samples()
        memory.add(rx, ry, rt, features)
        if use_device_replay:
            # gather the just-selected exemplars from the already-resident
            # task tensor (device-side; zero host traffic) and trim quotas
            if replay_mirror is None:
                from .cil.replay_gpu import DeviceReplayMirror
                replay_mirror = DeviceReplayMirror(device)
            replay_mirror.update(
                memory,
                task_images=train_loader.images[:train_loader.n_task],
                task_id=task_id)

        engine.detach()
        args.known_classes += args.increment_per_task  # before snapshot: the
        # checkpoint records the post-task state so resume starts task t+1
        ckpt = save_task_checkpoint(args.output_dir, task_id, model, memory,
                                    acc1s, args, optimizer, scheduler)
        if watchdog is not None and ckpt:
            watchdog.note_checkpoint(ckpt)

    avg_inc_acc = sum(acc1s) / len(acc1s) if acc1s else 0.0
    print(f"average incremental accuracy = {avg_inc_acc:.5f}")
    return acc1s
