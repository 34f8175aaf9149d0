"""Sequence-to-vector encoders for the classic NLP pipelines.

Reference behavior: paddlenlp/seq2vec/encoder.py (BoW/CNN/GRU/LSTM/RNN/TCN
encoders producing one vector per sequence).  Thin torch modules — these
feed the small classification heads, not the LLM path.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _masked(x: torch.Tensor, mask: Optional[torch.Tensor]):
    if mask is None:
        return x
    return x * mask.unsqueeze(-1).to(x.dtype)


class BoWEncoder(nn.Module):
    """Sum of (masked) token embeddings (reference :23)."""

    def __init__(self, emb_dim: int):
        super().__init__()
        self._emb_dim = emb_dim

    def get_input_dim(self):
        return self._emb_dim

    def get_output_dim(self):
        return self._emb_dim

    def forward(self, inputs, mask=None):
        return _masked(inputs, mask).sum(dim=1)


class CNNEncoder(nn.Module):
    """Parallel 1-D convs + max-over-time pooling (reference :125)."""

    def __init__(self, emb_dim: int, num_filter: int,
                 ngram_filter_sizes: List[int] = (2, 3, 4, 5),
                 conv_layer_activation=None, output_dim: Optional[int] = None):
        super().__init__()
        self._emb_dim = emb_dim
        self.convs = nn.ModuleList([
            nn.Conv1d(emb_dim, num_filter, k) for k in ngram_filter_sizes])
        self.act = conv_layer_activation or torch.tanh
        maxpool_dim = num_filter * len(ngram_filter_sizes)
        self.projection = (nn.Linear(maxpool_dim, output_dim)
                           if output_dim else None)
        self._output_dim = output_dim or maxpool_dim

    def get_input_dim(self):
        return self._emb_dim

    def get_output_dim(self):
        return self._output_dim

    def forward(self, inputs, mask=None):
        x = _masked(inputs, mask).transpose(1, 2)  # [B, E, S]
        pooled = [self.act(conv(x)).amax(dim=2) for conv in self.convs]
        out = torch.cat(pooled, dim=1)
        if self.projection is not None:
            out = self.projection(out)
        return out


class _RecurrentEncoder(nn.Module):
    rnn_cls = nn.RNN

    def __init__(self, input_size: int, hidden_size: int, num_layers: int = 1,
                 direction: str = "forward", dropout: float = 0.0,
                 pooling_type: Optional[str] = None):
        super().__init__()
        self._input_size = input_size
        self._hidden_size = hidden_size
        self.bidirectional = direction in ("bidirect", "bidirectional")
        self.pooling_type = pooling_type
        self.rnn = self.rnn_cls(
            input_size, hidden_size, num_layers=num_layers, batch_first=True,
            bidirectional=self.bidirectional,
            dropout=dropout if num_layers > 1 else 0.0)

    def get_input_dim(self):
        return self._input_size

    def get_output_dim(self):
        return self._hidden_size * (2 if self.bidirectional else 1)

    def _last_state(self, state):
        if isinstance(state, tuple):  # LSTM (h, c)
            state = state[0]
        if self.bidirectional:
            return torch.cat([state[-2], state[-1]], dim=-1)
        return state[-1]

    def forward(self, inputs, sequence_length=None, mask=None):
        out, state = self.rnn(inputs)
        if self.pooling_type is None:
            return self._last_state(state)
        out = _masked(out, mask)
        if self.pooling_type == "sum":
            return out.sum(dim=1)
        if self.pooling_type == "mean":
            n = (mask.sum(dim=1, keepdim=True).clamp(min=1).to(out.dtype)
                 if mask is not None else out.shape[1])
            return out.sum(dim=1) / n
        if self.pooling_type == "max":
            return out.amax(dim=1)
        raise ValueError(f"unknown pooling_type {self.pooling_type!r}")


class RNNEncoder(_RecurrentEncoder):
    rnn_cls = nn.RNN


class GRUEncoder(_RecurrentEncoder):
    rnn_cls = nn.GRU


class LSTMEncoder(_RecurrentEncoder):
    rnn_cls = nn.LSTM


class _TemporalBlock(nn.Module):
    def __init__(self, in_ch, out_ch, kernel_size, dilation, dropout):
        super().__init__()
        self.pad = (kernel_size - 1) * dilation
        self.conv1 = nn.Conv1d(in_ch, out_ch, kernel_size, dilation=dilation)
        self.conv2 = nn.Conv1d(out_ch, out_ch, kernel_size, dilation=dilation)
        self.dropout = nn.Dropout(dropout)
        self.downsample = (nn.Conv1d(in_ch, out_ch, 1)
                           if in_ch != out_ch else None)

    def _causal(self, conv, x):
        return conv(F.pad(x, (self.pad, 0)))

    def forward(self, x):  # [B, C, S]
        h = self.dropout(F.relu(self._causal(self.conv1, x)))
        h = self.dropout(F.relu(self._causal(self.conv2, h)))
        res = x if self.downsample is None else self.downsample(x)
        return F.relu(h + res)


class TCNEncoder(nn.Module):
    """Temporal convolutional network; output = last time step (reference :915)."""

    def __init__(self, input_size: int, num_channels: List[int],
                 kernel_size: int = 2, dropout: float = 0.2):
        super().__init__()
        layers = []
        for i, ch in enumerate(num_channels):
            in_ch = input_size if i == 0 else num_channels[i - 1]
            layers.append(_TemporalBlock(in_ch, ch, kernel_size, 2 ** i, dropout))
        self.network = nn.Sequential(*layers)
        self._input_size = input_size
        self._output_dim = num_channels[-1]

    def get_input_dim(self):
        return self._input_size

    def get_output_dim(self):
        return self._output_dim

    def forward(self, inputs):  # [B, S, E]
        return self.network(inputs.transpose(1, 2))[:, :, -1]
