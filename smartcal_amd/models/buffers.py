"""TrainingBuffer — (metadata, hint) pair store for distillation.

Parity with `demixing_rl/training_buffer.py:5-51`: x_/y_ arrays, ring
counter, pickle checkpoint to `databuffer.npy`.
"""

from __future__ import annotations

import pickle

import numpy as np

__all__ = ["TrainingBuffer"]


class TrainingBuffer:
    def __init__(self, max_size: int, n_input: int, n_output: int):
        self.mem_size = int(max_size)
        self.mem_cntr = 0
        self.x_ = np.zeros((self.mem_size, n_input), dtype=np.float32)
        self.y_ = np.zeros((self.mem_size, n_output), dtype=np.float32)
        self.filename = "databuffer.npy"

    def store(self, x, y):
        i = self.mem_cntr % self.mem_size
        self.x_[i] = x
        self.y_[i] = y
        self.mem_cntr += 1

    def sample(self, batch_size: int):
        filled = min(self.mem_cntr, self.mem_size)
        idx = np.random.choice(filled, batch_size,
                               replace=filled < batch_size)
        return self.x_[idx], self.y_[idx]

    def save_checkpoint(self, filename=None):
        with open(filename or self.filename, "wb") as f:
            pickle.dump({"mem_size": self.mem_size,
                         "mem_cntr": self.mem_cntr,
                         "x_": self.x_, "y_": self.y_}, f)

    def load_checkpoint(self, filename=None):
        with open(filename or self.filename, "rb") as f:
            d = pickle.load(f)
        self.mem_size = d["mem_size"]
        self.mem_cntr = d["mem_cntr"]
        self.x_ = d["x_"]
        self.y_ = d["y_"]
