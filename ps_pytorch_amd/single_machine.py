"""Single-machine CLI (reference parity: src/single_machine.py) —
BASELINE config 1: LeNet on MNIST, runs without a GPU."""
from __future__ import annotations


from .config import JobConfig, parse_args
from .data import prepare_data
from .trainer import NNTrainer


def main(argv=None) -> None:
    args = parse_args(argv)
    cfg = JobConfig.from_args(args)
    trainer = NNTrainer(cfg)
    trainer.build_model()
    train_loader, test_loader = prepare_data(
        cfg, device=trainer.device, dtype=trainer.compute_dtype)
    trainer.train_and_validate(train_loader, test_loader)


if __name__ == '__main__':
    main()
