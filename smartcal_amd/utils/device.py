"""Device selection + seeding.

Replaces the reference's per-module ``use_cuda=True; mydevice=...`` idiom
(reference ``elasticnet/enet_sac.py:11-15`` and ~20 other copies) with one
function. On a ROCm build ``torch.device('cuda')`` IS the MI355X HIP device.
"""

from __future__ import annotations

import os
import random

import numpy as np
import torch


def default_device() -> torch.device:
    """GPU if available (MI355X under ROCm), else CPU."""
    if torch.cuda.is_available():
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        return torch.device("cuda", local_rank % torch.cuda.device_count())
    return torch.device("cpu")


def seed_everything(seed: int) -> None:
    """Seed python / numpy / torch (reference plumbs np+torch seeds through
    every argparse main, e.g. ``elasticnet/main_sac.py:25-26``)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
