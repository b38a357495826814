#!/usr/bin/env python3
"""Evaluation CLI: validate a checkpoint (optionally its EMA weights).

    python eval.py app:apps/mobilenet_v3_large.yml train.resume=ckpt.pt
"""

import sys

from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.parallel import distributed as udist
from yet_another_mobilenet_series_amd.utils.config import setup_config


def main(argv=None) -> None:
    cfg = setup_config(argv if argv is not None else sys.argv[1:])
    trainer = Trainer(cfg)
    try:
        trainer.validate(epoch=trainer.start_epoch,
                         use_ema=cfg.val.use_ema and trainer.ema is not None)
    finally:
        udist.destroy()


if __name__ == "__main__":
    main()
