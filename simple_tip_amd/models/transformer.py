"""IMDB text classifier: 1-block transformer.

Mirrors the reference architecture (case_study_imdb.py:48-182): token+position
embedding (dim 32), one transformer block (2 heads, ff 32, post-LN, dropout
0.1), global average pooling, Dense(20, relu), Dense(2). Layer indices match
the keras functional model's ``model.layers`` (InputLayer at 0):
0 input(identity), 1 embedding, 2 transformer block, 3 GAP, 4 dropout,
5 dense20+relu (the SA tap, 20 ATs — reference SA_ACTIVATION_LAYERS=[5]),
6 dropout, 7 dense2. NC taps: [3, 5] (the reference's int-valued entries;
its tuple entries are dead config — see SURVEY.md §2.1).
"""

import torch
import torch.nn as nn

from .base import TapModel


class TokenAndPositionEmbedding(nn.Module):
    def __init__(self, maxlen: int, vocab_size: int, embed_dim: int):
        super().__init__()
        self.token_emb = nn.Embedding(vocab_size, embed_dim)
        self.pos_emb = nn.Embedding(maxlen, embed_dim)

    def forward(self, x):
        positions = torch.arange(x.shape[-1], device=x.device)
        return self.token_emb(x) + self.pos_emb(positions)


class TransformerBlock(nn.Module):
    def __init__(self, embed_dim: int, num_heads: int, ff_dim: int, rate: float = 0.1):
        super().__init__()
        self.att = nn.MultiheadAttention(embed_dim, num_heads, batch_first=True)
        self.ffn = nn.Sequential(
            nn.Linear(embed_dim, ff_dim), nn.ReLU(), nn.Linear(ff_dim, embed_dim)
        )
        self.layernorm1 = nn.LayerNorm(embed_dim, eps=1e-6)
        self.layernorm2 = nn.LayerNorm(embed_dim, eps=1e-6)
        self.dropout1 = nn.Dropout(rate)
        self.dropout2 = nn.Dropout(rate)

    def forward(self, x):
        attn_out, _ = self.att(x, x, x, need_weights=False)
        x = self.layernorm1(x + self.dropout1(attn_out))
        ffn_out = self.ffn(x)
        return self.layernorm2(x + self.dropout2(ffn_out))


class _GlobalAveragePooling1D(nn.Module):
    def forward(self, x):
        return x.mean(dim=1)


class ImdbTransformer(TapModel):
    """1-block transformer binary sentiment classifier (seq_len 100)."""

    num_classes = 2
    input_shape = (100,)
    sa_layers = [5]
    nc_layers = [3, 5]

    def __init__(self, vocab_size: int = 2000, maxlen: int = 100):
        super().__init__()
        self.layers = nn.ModuleList(
            [
                nn.Identity(),  # keras InputLayer placeholder (index parity)
                TokenAndPositionEmbedding(maxlen, vocab_size, 32),
                TransformerBlock(32, 2, 32),
                _GlobalAveragePooling1D(),
                nn.Dropout(0.1),
                nn.Sequential(nn.Linear(32, 20), nn.ReLU()),
                nn.Dropout(0.1),
                nn.Linear(20, 2),
            ]
        )
