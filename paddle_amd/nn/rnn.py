"""RNN layers (reference: python/paddle/nn/layer/rnn.py -- SimpleRNN,
LSTM, GRU + cells).  Compute runs on torch's fused RNN kernels (MIOpen
path on GPU); the paddle API shape (batch-major [B, T, C] default,
(h, c) state tuples, num_layers/direction) is preserved."""
from __future__ import annotations

import torch

from .layer import Layer


class _RNNBase(Layer):
    _mode = "RNN_TANH"

    def __init__(self, input_size, hidden_size, num_layers=1, direction="forward",
                 time_major=False, dropout=0.0, weight_ih_attr=None,
                 weight_hh_attr=None, bias_ih_attr=None, bias_hh_attr=None,
                 activation="tanh", name=None):
        super().__init__()
        bidirectional = direction in ("bidirect", "bidirectional")
        kind = {"RNN_TANH": "RNN", "RNN_RELU": "RNN", "LSTM": "LSTM", "GRU": "GRU"}[self._mode]
        kwargs = dict(input_size=input_size, hidden_size=hidden_size,
                      num_layers=num_layers, batch_first=not time_major,
                      dropout=dropout, bidirectional=bidirectional)
        if kind == "RNN":
            kwargs["nonlinearity"] = "tanh" if self._mode == "RNN_TANH" else "relu"
        self._rnn = getattr(torch.nn, kind)(**kwargs)
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.num_layers = num_layers
        self.num_directions = 2 if bidirectional else 1
        self.time_major = time_major

    def forward(self, inputs, initial_states=None, sequence_length=None):
        out, st = self._rnn(inputs, initial_states)
        return out, st


class SimpleRNN(_RNNBase):
    _mode = "RNN_TANH"


class LSTM(_RNNBase):
    _mode = "LSTM"


class GRU(_RNNBase):
    _mode = "GRU"


class LSTMCell(Layer):
    def __init__(self, input_size, hidden_size, weight_ih_attr=None,
                 weight_hh_attr=None, bias_ih_attr=None, bias_hh_attr=None, name=None):
        super().__init__()
        self._cell = torch.nn.LSTMCell(input_size, hidden_size)
        self.hidden_size = hidden_size

    def forward(self, inputs, states=None):
        h, c = self._cell(inputs, states)
        return h, (h, c)


class GRUCell(Layer):
    def __init__(self, input_size, hidden_size, **kw):
        super().__init__()
        self._cell = torch.nn.GRUCell(input_size, hidden_size)

    def forward(self, inputs, states=None):
        h = self._cell(inputs, states)
        return h, h


class SimpleRNNCell(Layer):
    def __init__(self, input_size, hidden_size, activation="tanh", **kw):
        super().__init__()
        self._cell = torch.nn.RNNCell(input_size, hidden_size,
                                      nonlinearity=activation)

    def forward(self, inputs, states=None):
        h = self._cell(inputs, states)
        return h, h
