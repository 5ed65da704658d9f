"""Small encoder Transformer — BASELINE.json config 5 (LOCO ablation study).

Built from named child modules so the LOCO ablator can drop layers/groups by
module name (the PyTorch analog of the reference's Keras-JSON layer surgery,
/root/reference/maggy/ablation/ablator/loco.py:99-136).
"""
import torch
import torch.nn as nn


class SmallTransformer(nn.Module):
    def __init__(self, vocab_size=1000, dim=128, n_heads=4, n_layers=4,
                 ffn_mult=4, num_classes=2, max_seq_len=128, dropout=0.0):
        super().__init__()
        self.embed = nn.Embedding(vocab_size, dim)
        self.pos_embed = nn.Embedding(max_seq_len, dim)
        self.blocks = nn.ModuleList(
            nn.TransformerEncoderLayer(
                d_model=dim, nhead=n_heads, dim_feedforward=dim * ffn_mult,
                dropout=dropout, batch_first=True, norm_first=True)
            for _ in range(n_layers))
        self.norm = nn.LayerNorm(dim)
        self.head = nn.Linear(dim, num_classes)

    def forward(self, tokens):
        T = tokens.shape[1]
        pos = torch.arange(T, device=tokens.device)
        x = self.embed(tokens) + self.pos_embed(pos)[None]
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x).mean(dim=1)
        return self.head(x)
