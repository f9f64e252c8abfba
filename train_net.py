"""Training entry point (reference train_net.py:1-13 parity).

Usage:
    python -m torch.distributed.run --nproc-per-node 8 --master-addr 127.0.0.1 \
        train_net.py --cfg config/resnet50.yaml [KEY VALUE ...]
"""

import distribuuuu_amd.trainer as trainer
from distribuuuu_amd.config import cfg, load_cfg_fom_args


def main():
    load_cfg_fom_args("Train a classification model.")
    cfg.freeze()
    trainer.train_model()


if __name__ == "__main__":
    main()
