# -*- coding: utf-8 -*-
"""Char-GRU for Shakespeare (parity with reference `nonconvex/rnn.py:8-41`).

Persistent hidden state carried across batches like the reference
(`rnn.py:27-35`): the hidden state is detached each forward and re-initialized
whenever the batch size changes.
"""
import torch.nn as nn


class CharGRU(nn.Module):
    def __init__(self, dataset, input_size, hidden_size, output_size,
                 batch_size, n_layers=1):
        super().__init__()
        if dataset not in ('shakespeare',):
            raise NotImplementedError(dataset)
        self.dataset = dataset
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.output_size = output_size
        self.n_layers = n_layers
        self.batch_size = batch_size
        self.encoder = nn.Embedding(input_size, hidden_size)
        self.gru = nn.GRU(hidden_size, hidden_size, n_layers, batch_first=True)
        self.decoder = nn.Linear(hidden_size, output_size)
        self.hidden = None
        self.init_hidden(batch_size)

    def init_hidden(self, batch_size=None):
        if batch_size is None:
            batch_size = self.batch_size
        weight = next(self.parameters())
        self.hidden = weight.new_zeros(self.n_layers, batch_size,
                                       self.hidden_size)

    def forward(self, x):
        if self.hidden is None or self.hidden.size(1) != x.size(0):
            self.init_hidden(x.size(0))
        if self.hidden.device != x.device:
            self.hidden = self.hidden.to(x.device)
        emb = self.encoder(x)
        out, h = self.gru(emb, self.hidden.detach())
        self.hidden = h.detach()
        out = self.decoder(out)
        # (B, C, T) for CrossEntropyLoss over characters
        return out.permute(0, 2, 1)


def rnn(args):
    return CharGRU(dataset=args.data, input_size=args.vocab_size,
                   hidden_size=args.rnn_hidden_size,
                   output_size=args.vocab_size, batch_size=args.batch_size)
