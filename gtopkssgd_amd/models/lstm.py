"""2-layer LSTM language model for PTB.

Reference parity: `upstream:lstmpy.py` — embedding,
2-layer LSTM, decoder, hidden-state detach (`repackage_hidden`).  PTB
vocabulary is 10000, standard hidden size 1500 in the large variant;
the reference family trains hidden≈1500, BPTT≈35.
"""
from typing import Tuple

import torch
import torch.nn as nn


def repackage_hidden(h):
    """Detach hidden state from the graph (truncated BPTT)."""
    if isinstance(h, torch.Tensor):
        return h.detach()
    return tuple(repackage_hidden(v) for v in h)


class LSTMModel(nn.Module):
    def __init__(self, vocab_size: int = 10000, emb_size: int = 1500,
                 hidden_size: int = 1500, num_layers: int = 2,
                 dropout: float = 0.65, tie_weights: bool = True):
        super().__init__()
        self.drop = nn.Dropout(dropout)
        self.encoder = nn.Embedding(vocab_size, emb_size)
        self.rnn = nn.LSTM(emb_size, hidden_size, num_layers,
                           dropout=dropout, batch_first=False)
        self.decoder = nn.Linear(hidden_size, vocab_size)
        if tie_weights:
            assert emb_size == hidden_size
            self.decoder.weight = self.encoder.weight
        self.hidden_size = hidden_size
        self.num_layers = num_layers
        self.vocab_size = vocab_size
        init = 0.1
        nn.init.uniform_(self.encoder.weight, -init, init)
        nn.init.zeros_(self.decoder.bias)

    def init_hidden(self, batch_size: int, device=None
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
        w = next(self.parameters())
        shape = (self.num_layers, batch_size, self.hidden_size)
        return (w.new_zeros(shape), w.new_zeros(shape))

    def forward(self, tokens, hidden):
        # tokens: [seq_len, batch]
        emb = self.drop(self.encoder(tokens))
        out, hidden = self.rnn(emb, hidden)
        out = self.drop(out)
        logits = self.decoder(out)  # [seq, batch, vocab]
        return logits, hidden


def lstm_ptb(vocab_size=10000, hidden=1500):
    return LSTMModel(vocab_size=vocab_size, emb_size=hidden,
                     hidden_size=hidden)
