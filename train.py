#!/usr/bin/env python3
"""Training CLI (mirrors the reference's UX):

    torchrun --nproc-per-node 8 train.py app:apps/mobilenet_v3_large.yml \
        optim.lr=0.8 train.epochs=150
"""

import sys

from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.parallel import distributed as udist
from yet_another_mobilenet_series_amd.utils.config import setup_config


def main(argv=None) -> None:
    cfg = setup_config(argv if argv is not None else sys.argv[1:])
    trainer = Trainer(cfg)
    try:
        trainer.fit()
    finally:
        udist.destroy()


if __name__ == "__main__":
    main()
