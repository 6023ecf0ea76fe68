"""Seeding (reference: d9d/internals/determinism/seed.py:11-62).

Base seed + pp-rank offset: pipeline stages draw DIFFERENT dropout/init
streams; data-parallel replicas share one so replicated init matches.
"""

import os
import random

import numpy as np
import torch


def set_seeds(base_seed: int, pp_rank: int = 0) -> None:
    seed = base_seed + pp_rank
    torch.manual_seed(seed)
    random.seed(seed)
    np.random.seed(seed % (2**32))
    os.environ["PYTHONHASHSEED"] = str(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
