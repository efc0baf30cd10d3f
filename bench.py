"""Flagship benchmark: ResNet-18 / CIFAR-10-shaped synthetic data, bf16.

Measures the BASELINE.json headline metric — images/sec (whole node) for the
synchronous PS engine — at N GPUs (N=1: the single-machine engine; N>1:
1 PS on rank 0 + N-1 workers, batch 1024 per worker, weak scaling).
Synthetic data / random-init weights per BASELINE.json (no network for
datasets); dtype bf16 (compute) with an f32 master; the timed region
includes forward + backward + comm + the fused optimizer step every
iteration. Expected scaling shape: rank 0 computes no samples, so
value(N) ~ (N-1) x per-worker throughput (docs/RCCL_READINESS.md).

Single GPU:      python bench.py --gpus 1 --steps 30 --warmup 5
Multi GPU (driver launches):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=30)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--batch-size', type=int, default=1024,
                   help='per-worker batch size')
    p.add_argument('--network', type=str, default='ResNet18')
    p.add_argument('--dataset', type=str, default='Cifar10')
    p.add_argument('--compress-grad', type=str, default='compress')
    p.add_argument('--bucket-mb', type=float, default=4.0)
    p.add_argument('--no-overlap', action='store_true')
    p.add_argument('--aggregation', type=str, default='collective',
                   help='collective | gather (PS engine fan-in mode)')
    p.add_argument('--engine', type=str, default='ps',
                   help='ps | allreduce (N>1 only)')
    return p.parse_args()


def make_batches(n_batches, bs, device, dtype, seed=1234,
                 dataset='Cifar10'):
    from ps_pytorch_amd.config import input_shape_of, num_classes_of
    shape = input_shape_of(dataset)
    ncls = num_classes_of(dataset)
    g = torch.Generator(device='cpu').manual_seed(seed)
    xs, ys = [], []
    for _ in range(n_batches):
        x = torch.randn(bs, *shape, generator=g).to(device, dtype)
        if device.type == 'cuda':
            x = x.contiguous(memory_format=torch.channels_last)
        xs.append(x)
        ys.append(torch.randint(0, ncls, (bs,), generator=g).to(device))
    return xs, ys


def main():
    args = parse()
    from ps_pytorch_amd.config import JobConfig
    from ps_pytorch_amd.parallel.transport import init_distributed

    from ps_pytorch_amd.config import num_classes_of, input_shape_of
    ncls = num_classes_of(args.dataset)
    ishape = input_shape_of(args.dataset)
    world = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    n_gpus = max(world, 1)
    use_cuda = torch.cuda.is_available()
    cfg = JobConfig(network=args.network, dataset=args.dataset,
                    batch_size=args.batch_size, lr=0.1, momentum=0.9,
                    max_steps=args.steps + args.warmup,
                    compress_grad=args.compress_grad,
                    bucket_mb=args.bucket_mb, overlap=not args.no_overlap,
                    aggregation=args.aggregation, engine=args.engine,
                    enable_gpu=use_cuda, eval_freq=10 ** 9,
                    train_dir='/tmp/ps_bench_models')

    if world > 1:
        import torch.distributed as dist
        env = init_distributed()
        device = env['device']
        # timing barriers run on a HOST-SIDE gloo group: a barrier on the
        # main (RCCL) communicator would be a device collective that the PS
        # enqueues AFTER its pipelined tail broadcasts while workers
        # enqueue it BEFORE their matching receives — mismatched collective
        # order deadlocks NCCL (gloo tolerates it, which is why CPU tests
        # can't catch this one)
        barrier_grp = dist.new_group(backend='gloo')
        from ps_pytorch_amd.parallel.ps import ParameterServer
        from ps_pytorch_amd.parallel.worker import DistributedWorker
        if args.engine == 'allreduce':
            from ps_pytorch_amd.parallel.allreduce import AllReduceTrainer
            role = AllReduceTrainer(cfg, rank, world, device)
            role.build_model(ncls)
            xs, ys = make_batches(8, args.batch_size, device,
                                  role.compute_dtype, seed=1234 + rank,
                                  dataset=args.dataset)
            it = [0]

            def step():
                i = it[0] % len(xs)
                it[0] += 1
                role.train_step(xs[i], ys[i])
        elif rank == 0:
            role = ParameterServer(cfg, rank, world, device)
            role.build_model(ncls)
            step = role.step
        else:
            role = DistributedWorker(cfg, rank, world, device)
            role.build_model(ncls)
            xs, ys = make_batches(8, args.batch_size, device,
                                  role.compute_dtype, seed=1234 + rank,
                                  dataset=args.dataset)
            it = [0]

            def step():
                i = it[0] % len(xs)
                it[0] += 1
                role.train_step(xs[i], ys[i])

        for _ in range(args.warmup):
            step()
        if use_cuda:
            torch.cuda.synchronize()
        dist.barrier(group=barrier_grp)
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(args.steps):
            step()
        if use_cuda:
            torch.cuda.synchronize()
        dist.barrier(group=barrier_grp)
        if use_cuda:
            torch.cuda.synchronize()
        elapsed = time.time() - t0
        # max over ranks
        et = torch.tensor([elapsed], dtype=torch.float64,
                          device=device if env['backend'] == 'nccl' else 'cpu')
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        elapsed = float(et)
        if args.engine == 'allreduce':
            n_workers = world
            parallelism = f"allreduce-dp ({world} ranks)"
        else:
            n_workers = world - 1
            parallelism = f"ps-dp (1 PS + {n_workers} workers)"
        imgs = n_workers * args.batch_size * args.steps
        global_batch = n_workers * args.batch_size
        # config echo describes what actually ran on the wire
        cfg_extra = {"compress_grad": (args.compress_grad
                                       if args.engine == 'ps' else 'None'),
                     "aggregation": (args.aggregation
                                     if args.engine == 'ps' else 'allreduce'),
                     "overlap": not args.no_overlap}
    else:
        from ps_pytorch_amd.trainer import NNTrainer
        device = torch.device('cuda', 0) if use_cuda else torch.device('cpu')
        tr = NNTrainer(cfg, device=device)
        tr.build_model(ncls)
        xs, ys = make_batches(8, args.batch_size, device, tr.compute_dtype,
                              dataset=args.dataset)
        graphed = use_cuda and tr.enable_graph(xs[0], ys[0])
        step_fn = tr.graph_step if graphed else tr.train_step
        for i in range(args.warmup):
            step_fn(xs[i % len(xs)], ys[i % len(ys)])
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.time()
        for i in range(args.steps):
            step_fn(xs[i % len(xs)], ys[i % len(ys)])
        if use_cuda:
            torch.cuda.synchronize()
        elapsed = time.time() - t0
        imgs = args.batch_size * args.steps
        parallelism = "single-gpu"
        global_batch = args.batch_size
        # honesty: the N=1 path is the single-machine engine — no PS, no
        # comm, no compression ran inside the timed region; say exactly that
        # instead of echoing distributed-only flags
        cfg_extra = {"engine": "single-machine trainer"
                               + (" (hipGraph-captured step)" if graphed
                                  else " (eager)"),
                     "compress_grad": "n/a (no comm at N=1)",
                     "overlap": "n/a (no comm at N=1)"}

    if world > 1:
        # orderly teardown: exiting with live comm threads makes gloo's C++
        # layer abort sporadically at process exit ("terminate called
        # without an active exception") — torchrun then reports the whole
        # run failed even though the metric was printed
        import torch.distributed as dist
        dist.barrier(group=barrier_grp)
        dist.destroy_process_group()

    if rank == 0:
        value = imgs / elapsed
        dtype = 'bf16' if use_cuda else 'fp32'
        print(json.dumps({
            "metric": (f"images/sec (whole node), "
                       f"{'ResNet-18' if args.network == 'ResNet18' else args.network}"
                       f"/{'CIFAR-10' if args.dataset == 'Cifar10' else args.dataset}"
                       " synchronous PS training"),
            "value": value,
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic (random-init weights, GPU-resident random batches)",
            "config": {"model": args.network, "global_batch": global_batch,
                       "input": "x".join(str(d) for d in ishape),
                       "per_worker_batch": args.batch_size,
                       "parallelism": parallelism, **cfg_extra},
        }))


if __name__ == '__main__':
    main()
