#!/usr/bin/env python3
"""Training entry point (parity: ESR:train_ours_cnt_seq.py:742-832).

Single GPU:
    python train.py -c configs/train_synth_2x.yml
8 GPUs over RCCL/xGMI:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py -c configs/train_synth_2x.yml
"""

import argparse
import os
import random

import numpy as np
import torch

from esr_amd.config import ConfigParser
from esr_amd.engine import build_training
from esr_amd.parallel import get_rank, init_distributed, setup_rank0_print


def init_seeds(seed=0):
    os.environ["PYTHONHASHSEED"] = str(seed)
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def main():
    parser = argparse.ArgumentParser(description="esr_amd trainer")
    parser.add_argument("-c", "--config", required=True, type=str)
    parser.add_argument("-id", "--runid", default=None, type=str)
    parser.add_argument("-seed", "--seed", default=123, type=int)
    parser.add_argument("-r", "--resume", default=None, type=str)
    parser.add_argument("--reset", action="store_true",
                        help="on resume, reset trainer counters")
    args = parser.parse_args()

    local_rank = init_distributed()
    setup_rank0_print()
    init_seeds(args.seed + get_rank())

    config_parser = ConfigParser.from_file(
        args.config, run_id=args.runid, make_dirs=get_rank() == 0, args=args)
    device = torch.device(f"cuda:{local_rank}") if torch.cuda.is_available() \
        else torch.device("cpu")
    logger = config_parser.get_logger("train")

    trainer = build_training(config_parser, device, logger,
                             resume=args.resume, reset=args.reset)
    trainer.train()


if __name__ == "__main__":
    main()
