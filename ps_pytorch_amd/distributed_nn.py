"""Distributed CLI / role mux (reference parity: src/distributed_nn.py).

Launch (one rank per GPU over RCCL; replaces `mpirun -n N`):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 \
      -m ps_pytorch_amd.distributed_nn --network=ResNet18 --dataset=Cifar10 ...

Rank 0 becomes the PS; ranks 1..N-1 become workers (ref :109-125).
"""
from __future__ import annotations


from .config import JobConfig, parse_args, num_classes_of
from .data import prepare_data
from .parallel.ps import ParameterServer
from .parallel.transport import init_distributed
from .parallel.worker import DistributedWorker
from .utils.logging import get_logger

logger = get_logger('ps_pytorch_amd.main')


def main(argv=None) -> None:
    args = parse_args(argv)
    cfg = JobConfig.from_args(args)
    env = init_distributed()
    rank, world, device = env['rank'], env['world'], env['device']
    if world < 2:
        raise SystemExit("distributed_nn needs world_size >= 2 "
                         "(1 PS + >=1 worker); use single_machine.py for 1 rank")
    nc = num_classes_of(cfg.dataset)
    if cfg.engine == 'allreduce':
        from .parallel.allreduce import AllReduceTrainer
        tr = AllReduceTrainer(cfg, rank, world, device)
        tr.build_model(nc)
        train_loader, _ = prepare_data(cfg, rank=rank, num_shards=world,
                                       device=device, dtype=tr.compute_dtype)
        step = 0
        while step < cfg.max_steps:
            for data, target in train_loader:
                if step >= cfg.max_steps:
                    break
                loss = tr.train_step(data, target)
                step += 1
                if rank == 0 and step % cfg.log_interval == 0:
                    logger.info('AllReduce step %d loss %.4f', step, float(loss))
        return
    if rank == 0:
        ps = ParameterServer(cfg, rank, world, device)
        ps.build_model(nc)
        logger.info('PS: model %s, %d params, %d buckets, wire=%s',
                    cfg.network, ps.flat.total, len(ps.flat.buckets),
                    ps.transport.wire_dtype)
        ps.start()
    else:
        worker = DistributedWorker(cfg, rank, world, device)
        worker.build_model(nc)
        train_loader, test_loader = prepare_data(
            cfg, rank=rank, num_shards=world - 1, device=device,
            dtype=worker.compute_dtype)
        worker.train(train_loader, test_loader)
    # orderly teardown: exiting with live comm threads sporadically aborts
    # in gloo's C++ layer ("terminate called without an active exception")
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
