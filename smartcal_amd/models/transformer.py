"""Transformer encoder classifier + supervised replay buffer.

Same model family as the reference (`calibration/transformer_models.py:
76-184`): fused-qkv multi-head attention computed per sample with the
head axis playing the token role (input is (batch, embed); the reference
reshapes to (batch, heads, 3·head_dim) at `:108` — attention mixes the
heads of one sample, there is no sequence axis), post-LN residual
blocks, sigmoid output head. Buffer layout matches
`transformer_models.ReplayBuffer:10-70` (x/y arrays, resize, pickle
checkpoints).
"""

from __future__ import annotations

import math
import pickle

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.linear import FusedLinear

__all__ = ["SupervisedBuffer", "TransformerEncoder",
           "scaled_dot_product", "MultiheadAttention", "EncoderBlock"]


class SupervisedBuffer:
    """(x, y) sample store with ring semantics + resize."""

    def __init__(self, max_size: int, input_shape, output_shape):
        self.mem_size = int(max_size)
        self.mem_cntr = 0
        self.x = np.zeros((self.mem_size, *input_shape), dtype=np.float32)
        self.y = np.zeros((self.mem_size, *output_shape), dtype=np.float32)
        self.filename = "simul_data.buffer"

    def store_data(self, x, y):
        i = self.mem_cntr % self.mem_size
        self.x[i] = x
        self.y[i] = y
        self.mem_cntr += 1

    def resize(self, newsize: int):
        assert newsize > self.mem_size
        xnew = np.zeros((newsize, *self.x.shape[1:]), dtype=np.float32)
        xnew[:self.mem_size] = self.x
        ynew = np.zeros((newsize, *self.y.shape[1:]), dtype=np.float32)
        ynew[:self.mem_size] = self.y
        self.x, self.y, self.mem_size = xnew, ynew, newsize

    def sample_minibatch(self, batch_size: int):
        filled = min(self.mem_cntr, self.mem_size)
        batch = np.random.choice(filled, batch_size, replace=False)
        return self.x[batch], self.y[batch]

    def save_checkpoint(self, filename=None):
        with open(filename or self.filename, "wb") as f:
            pickle.dump({"mem_size": self.mem_size,
                         "mem_cntr": self.mem_cntr,
                         "x": self.x, "y": self.y}, f)

    def load_checkpoint(self, filename=None):
        with open(filename or self.filename, "rb") as f:
            d = pickle.load(f)
        self.mem_size = d["mem_size"]
        self.mem_cntr = d["mem_cntr"]
        self.x = d["x"]
        self.y = d["y"]


def scaled_dot_product(q, k, v):
    """One fused HIP launch on GPU (ops/csrc/attention.hip: LDS-staged
    MFMA tiles + in-LDS softmax); torch composition on CPU (the oracle).
    Same contract as the reference's (`transformer_models.py:76-83`)."""
    from ..ops.attention import scaled_dot_product as _sdp
    return _sdp(q, k, v)


class MultiheadAttention(nn.Module):
    """Fused-qkv MHA over the head axis of one sample
    (`transformer_models.py:85-118`). Projections run on the
    hand-written fused-linear HIP kernels (ops/csrc/fused_linear.hip)."""

    def __init__(self, input_dim: int, embed_dim: int, num_heads: int):
        super().__init__()
        assert embed_dim % num_heads == 0
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.qkv_proj = FusedLinear(input_dim, 3 * embed_dim, ln=False,
                                    act="none")
        self.o_proj = FusedLinear(embed_dim, embed_dim, ln=False,
                                  act="none")
        nn.init.xavier_uniform_(self.qkv_proj.weight)
        self.qkv_proj.bias.data.fill_(0)
        nn.init.xavier_uniform_(self.o_proj.weight)
        self.o_proj.bias.data.fill_(0)

    def forward(self, x, return_attention: bool = False):
        batch_size, embed_dim = x.size()
        qkv = self.qkv_proj(x).reshape(batch_size, self.num_heads,
                                       3 * self.head_dim)
        q, k, v = qkv.chunk(3, dim=-1)
        values, attention = scaled_dot_product(q, k, v)
        o = self.o_proj(values.reshape(batch_size, embed_dim))
        return (o, attention) if return_attention else o


class EncoderBlock(nn.Module):
    """Post-LN residual encoder block (`transformer_models.py:121-150`)."""

    def __init__(self, input_dim, num_heads, dim_feedforward, dropout=0.0):
        super().__init__()
        self.self_attn = MultiheadAttention(input_dim, input_dim, num_heads)
        self.linear_net = nn.Sequential(
            FusedLinear(input_dim, dim_feedforward, ln=False, act="none",
                        init_scale=1.0 / input_dim ** 0.5),
            nn.Dropout(dropout),
            nn.ReLU(),
            FusedLinear(dim_feedforward, input_dim, ln=False, act="none",
                        init_scale=1.0 / dim_feedforward ** 0.5),
        )
        self.norm1 = nn.LayerNorm(input_dim)
        self.norm2 = nn.LayerNorm(input_dim)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x):
        x = self.norm1(x + self.dropout(self.self_attn(x)))
        x = self.norm2(x + self.dropout(self.linear_net(x)))
        return x


class TransformerEncoder(nn.Module):
    """input_net → blocks → output_net → sigmoid
    (`transformer_models.py:153-184`)."""

    def __init__(self, num_layers, input_dim, model_dim, num_classes,
                 num_heads, dropout=0.0):
        super().__init__()
        # input_dim can exceed the fused kernel\'s 640-column cap only on
        # the IN side (K is unbounded); FusedLinear handles any K
        self.input_net = nn.Sequential(
            nn.Dropout(dropout),
            FusedLinear(input_dim, model_dim, ln=False, act="none",
                        init_scale=1.0 / input_dim ** 0.5))
        self.layers = nn.ModuleList([
            EncoderBlock(model_dim, num_heads, model_dim, dropout)
            for _ in range(num_layers)])
        self.output_net = nn.Sequential(
            FusedLinear(model_dim, model_dim, ln=False, act="none",
                        init_scale=1.0 / model_dim ** 0.5),
            nn.LayerNorm(model_dim),
            nn.ReLU(),
            nn.Dropout(dropout),
            FusedLinear(model_dim, num_classes, ln=False, act="none",
                        init_scale=1.0 / model_dim ** 0.5),
        )

    def forward(self, x):
        x = self.input_net(x)
        for l in self.layers:
            x = l(x)
        return torch.sigmoid(self.output_net(x))

    @torch.no_grad()
    def get_attention_maps(self, x):
        x = self.input_net(x)
        maps = []
        for l in self.layers:
            _, attn = l.self_attn(x, return_attention=True)
            maps.append(attn)
            x = l(x)
        return maps
