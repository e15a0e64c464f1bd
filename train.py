#!/usr/bin/env python3
"""Training entry point (reference-compatible CLI).

    python train.py -c config/default.yaml -n my_run [-p chkpt.pt]

Multi-GPU (one process per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py -c config.yaml -n run8
"""

from __future__ import annotations

import argparse
import random

import numpy as np
import torch

from melgan_amd.config import HParam, env_world, merge_default
from melgan_amd.data import create_dataloader
from melgan_amd.engine.trainer import Trainer
from melgan_amd.parallel.dp import init_distributed


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-c", "--config", required=True, help="yaml config file")
    ap.add_argument("-n", "--name", required=True, help="run name")
    ap.add_argument("-p", "--checkpoint_path", default=None,
                    help="checkpoint to resume from")
    ap.add_argument("--synthetic", action="store_true",
                    help="use synthetic mels (benchmarking; no dataset "
                         "needed)")
    ap.add_argument("--max-steps", type=int, default=None)
    args = ap.parse_args()

    hp = merge_default(HParam(args.config))
    rank, local_rank, world = env_world()
    if world > 1:
        init_distributed()

    seed = hp.train.seed + rank
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)

    trainer = Trainer(hp, args.name, rank=rank, world_size=world)
    if args.checkpoint_path:
        trainer.load(args.checkpoint_path)

    trainloader = create_dataloader(hp, train=True, synthetic=args.synthetic,
                                    rank=rank, world_size=world)
    try:
        valloader = create_dataloader(hp, train=False,
                                      synthetic=args.synthetic)
    except FileNotFoundError:
        valloader = None

    trainer.train(trainloader, valloader, max_steps=args.max_steps)
    trainer.save()

    # explicit clean exit on success: interpreter/HIP teardown after a GPU
    # training run can segfault (rc 139) AFTER all work completed — see
    # NOTES.md "exit-teardown segfault"; under torchrun that turns a
    # finished run into a reported failure
    import sys

    if world > 1 and torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    sys.stdout.flush()
    sys.stderr.flush()
    import os

    os._exit(0)


if __name__ == "__main__":
    main()
