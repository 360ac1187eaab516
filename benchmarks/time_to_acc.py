#!/usr/bin/env python3
"""Time-to-97%-accuracy benchmark (the second half of BASELINE.json's
headline metric) on the MNIST-shaped MLP.

There is no network access for real MNIST, so the dataset is a synthetic
10-class mixture in 784-dim (class template + gaussian noise, `--noise`
controls difficulty) — random-init weights, fp32, async-PS Hogwild.

Run like bench.py:
  python benchmarks/time_to_acc.py --gpus 1
  torchrun --nproc-per-node N benchmarks/time_to_acc.py --gpus N
"""
import argparse
import json
import os
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def make_templates(n_classes=10, dim=784, seed=77):
    """(task=mixture) class templates + isotropic noise — the round-1
    task.  NOTE: linearly separable; a linear probe alone reaches ~100%,
    so 97% here is a LOW bar.  Kept for comparison; the default task is
    the teacher-MLP one below."""
    return np.random.default_rng(seed).normal(0, 1, (n_classes, dim)) \
        .astype(np.float32)


def make_synthetic_mnist(rng, n, templates, noise=2.0):
    n_classes, dim = templates.shape
    labels = rng.integers(0, n_classes, n)
    x = templates[labels] + rng.normal(0, noise, (n, dim)).astype(np.float32)
    return x.astype(np.float32), labels.astype(np.int64)


def make_teacher(dim=784, latent=16, hidden=32, n_classes=10, seed=77):
    """(task=teacher, default) labels come from a fixed random tanh-MLP
    over a LOW-DIM latent, embedded into 784-d — an MNIST-like structure
    (low intrinsic dimension, nonlinear class boundaries, many nuisance
    pixel dims).  A linear probe in pixel space cannot represent the
    boundary (the harness prints its accuracy as evidence); margin
    filtering + pixel noise set how hard 97% is (round-1 verdict weak
    #4: the old mixture task was linearly separable)."""
    rng = np.random.default_rng(seed)
    P = rng.normal(0, 1.0, (latent, dim)).astype(np.float32)
    w1 = (rng.normal(0, 1.0, (latent, hidden)) * 4.0 /
          np.sqrt(latent)).astype(np.float32)
    w2 = rng.normal(0, 1.0 / np.sqrt(hidden),
                    (hidden, n_classes)).astype(np.float32)
    return P, w1, w2


def make_teacher_task(rng, n, teacher, noise=0.15, margin=0.7):
    P, w1, w2 = teacher
    zs, ys = [], []
    got = 0
    while got < n:
        z = rng.normal(0, 1, (2 * n, w1.shape[0])).astype(np.float32)
        logits = np.tanh(z @ w1) @ w2
        top2 = np.partition(logits, -2, axis=1)[:, -2:]
        keep = (top2[:, 1] - top2[:, 0]) > margin
        zs.append(z[keep])
        ys.append(logits[keep].argmax(axis=1))
        got += int(keep.sum())
    z = np.concatenate(zs)[:n]
    y = np.concatenate(ys)[:n]
    x = z @ P + rng.normal(0, noise * np.sqrt(P.shape[0]),
                           (n, P.shape[1])).astype(np.float32)
    return x.astype(np.float32), y.astype(np.int64)


def linear_probe_acc(x, y, ex, ey, n_classes=10, epochs=60, lr=0.5):
    """Multinomial logistic regression baseline (full-batch, CPU torch):
    quantifies how much of the task a LINEAR model already solves."""
    xt = torch.as_tensor(x)
    onehot = torch.eye(n_classes)[torch.as_tensor(y)]
    w = torch.zeros(x.shape[1], n_classes)
    b = torch.zeros(n_classes)
    for _ in range(epochs):
        p = torch.softmax(xt @ w + b, dim=1)
        g = (p - onehot) / x.shape[0]
        w -= lr * (xt.t() @ g)
        b -= lr * g.sum(0)
    pred = (torch.as_tensor(ex) @ w + b).argmax(dim=1)
    return float((pred == torch.as_tensor(ey)).float().mean())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--target", type=float, default=0.97)
    ap.add_argument("--mbs", type=int, default=300)
    ap.add_argument("--noise", type=float, default=None,
                    help="default: 5.0 (mixture) / 0.4 (teacher)")
    ap.add_argument("--task", default="teacher",
                    choices=["teacher", "mixture"])
    ap.add_argument("--lr", type=float, default=0.003)
    ap.add_argument("--round-steps", type=int, default=20)
    ap.add_argument("--max-rounds", type=int, default=400)
    ap.add_argument("--train-rows", type=int, default=60000)
    args = ap.parse_args()

    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.models import mnist_mlp
    from sparkflow_amd.parallel.dist_engine import build_engine
    from sparkflow_amd.parallel.ps import OptimizerSpec, TrainConfig
    from sparkflow_amd.utils.serialization import unpack_params

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))

    # seeded init: run-to-run tta variance was dominated by the random
    # start point, not the engine
    torch.manual_seed(4242)
    graph_json = build_graph(mnist_mlp)
    cfg = TrainConfig(iters=args.max_rounds * args.round_steps,
                      mini_batch=args.mbs, mini_stochastic_iters=1,
                      shuffle=False, seed=7)  # seeded minibatch stream
    engine = build_engine(graph_json, OptimizerSpec("adam", args.lr), cfg)
    device = engine.device

    rng = np.random.default_rng(1000 + rank)
    ev_rng = np.random.default_rng(999)
    if args.task == "teacher":
        noise = 0.15 if args.noise is None else args.noise
        teacher = make_teacher()
        x, y = make_teacher_task(rng, args.train_rows, teacher, noise)
        if rank == 0:
            ex, ey = make_teacher_task(ev_rng, 2000, teacher, noise)
    else:
        noise = 5.0 if args.noise is None else args.noise
        templates = make_templates()
        x, y = make_synthetic_mnist(rng, args.train_rows, templates,
                                    noise=noise)
        if rank == 0:
            ex, ey = make_synthetic_mnist(ev_rng, 2000, templates,
                                          noise=noise)
    engine.worker.stage(x, y.astype(np.float32).reshape(-1, 1))

    probe_acc = linear_probe_acc(x, y, ex, ey) if rank == 0 else None
    if rank == 0:
        ex_t = torch.as_tensor(ex, device=device)
        ey_t = torch.as_tensor(ey, device=device)
        eval_module = engine.worker.module  # weights loaded from PS below

    # warmup builds communicators/graphs AND holds the GPU busy ~2 s so
    # the clocks ramp before the timer starts (first-process runs
    # otherwise measure the ramp, not the training)
    wt0 = time.perf_counter()
    engine.worker.run_steps(3, args.mbs)
    if device.type == "cuda":
        torch.cuda.synchronize()
        # clock-ramp burn on DUMMY work (extra training steps here would
        # corrupt the time-to-accuracy semantics)
        a = torch.randn(4096, 4096, device=device)
        while time.perf_counter() - wt0 < 2.0:
            a = a @ a * 1e-6
            torch.cuda.synchronize()
        del a
    engine.barrier()

    flag = torch.zeros(1, dtype=torch.int64)
    t0 = time.perf_counter()
    elapsed = float("nan")
    rounds = 0
    acc = 0.0
    for rounds in range(1, args.max_rounds + 1):
        engine.worker.run_steps(args.round_steps, args.mbs)
        if device.type == "cuda":
            torch.cuda.synchronize()
        engine.barrier()
        if rank == 0:
            unpack_params(eval_module, engine.ps.weights)
            with torch.no_grad():
                pred = engine.worker.module(ex_t).argmax(dim=1)
            acc = float((pred == ey_t).float().mean())
            flag.fill_(1 if acc >= args.target else 0)
        if world > 1:
            import torch.distributed as dist

            dist.broadcast(flag, src=0)
        if int(flag.item()) == 1:
            elapsed = time.perf_counter() - t0
            break

    if rank == 0:
        total_steps = rounds * args.round_steps
        print(json.dumps({
            "metric": "time_to_97pct_acc",
            "value": elapsed,
            "unit": "s",
            "n_gpus": world,
            "steps": total_steps,
            "warmup": 3,
            "ms_per_step": (elapsed / total_steps * 1000.0
                            if elapsed == elapsed else None),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "final_accuracy": acc,
            "task": args.task,
            "task_noise": noise,
            "linear_probe_accuracy": probe_acc,
            "config": {"model": "mnist_mlp_784_256_256_10",
                       "global_batch": args.mbs * world,
                       "seq_len": None,
                       "parallelism": f"async-ps-hogwild dp{world}"},
        }))

    engine.close()
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
