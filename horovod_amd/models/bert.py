"""BERT-large for the pretraining benchmark config (BASELINE.json config 4:
"BERT-large pretrain with fp16 grad compression + tensor-fusion autotune").

Own compact implementation on torch built-ins: nn.TransformerEncoder layers
use scaled_dot_product_attention, which routes to the ROCm fused-attention
backends on MI355X; bf16 autocast covers the matmuls via hipBLASLt.
"""
import torch
import torch.nn as nn


class BertConfig:
    def __init__(self, vocab_size=30522, hidden=1024, layers=24, heads=16,
                 intermediate=4096, max_seq=512, type_vocab=2, dropout=0.1):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.intermediate = intermediate
        self.max_seq = max_seq
        self.type_vocab = type_vocab
        self.dropout = dropout


BERT_LARGE = BertConfig()
BERT_BASE = BertConfig(hidden=768, layers=12, heads=12, intermediate=3072)


class BertForPretraining(nn.Module):
    """Embeddings + encoder + tied MLM head + NSP head."""

    def __init__(self, cfg=BERT_LARGE):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos_emb = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.type_emb = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.emb_norm = nn.LayerNorm(cfg.hidden)
        self.emb_drop = nn.Dropout(cfg.dropout)
        layer = nn.TransformerEncoderLayer(
            d_model=cfg.hidden, nhead=cfg.heads,
            dim_feedforward=cfg.intermediate, dropout=cfg.dropout,
            activation="gelu", batch_first=True, norm_first=False)
        self.encoder = nn.TransformerEncoder(layer, cfg.layers,
                                             enable_nested_tensor=False)
        self.mlm_transform = nn.Sequential(
            nn.Linear(cfg.hidden, cfg.hidden), nn.GELU(),
            nn.LayerNorm(cfg.hidden))
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.nsp = nn.Linear(cfg.hidden, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        x = self.tok_emb(input_ids) + self.pos_emb(pos)
        if token_type_ids is not None:
            x = x + self.type_emb(token_type_ids)
        x = self.emb_drop(self.emb_norm(x))
        pad_mask = None
        if attention_mask is not None:
            pad_mask = attention_mask == 0
        h = self.encoder(x, src_key_padding_mask=pad_mask)
        # tied MLM head (weight sharing with tok_emb, BERT-standard)
        mlm = self.mlm_transform(h) @ self.tok_emb.weight.t() + self.mlm_bias
        nsp = self.nsp(h[:, 0])
        return mlm, nsp


def bert_large():
    return BertForPretraining(BERT_LARGE)


def bert_base():
    return BertForPretraining(BERT_BASE)
