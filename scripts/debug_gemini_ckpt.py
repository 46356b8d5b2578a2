import faulthandler
import os
import sys

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def run(rank):
    faulthandler.enable()
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
    from test_checkpoint_io.test_gemini_ckpt import _run
    _run(rank, 2, 2977, "/tmp/gemckpt")


if __name__ == "__main__":
    os.makedirs("/tmp/gemckpt", exist_ok=True)
    mp.spawn(run, nprocs=2)
