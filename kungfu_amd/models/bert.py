"""BERT-base encoder (masked-LM head) for the SMA + gradient-noise-scale
benchmark config (BASELINE.json config 4).

Self-contained transformer encoder with BERT-base dimensions (L=12, H=768,
A=12, ~110M params); attention uses torch's fused SDPA, which lowers to the
ROCm flash-attention path on MI355X.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from kungfu_amd.ops.fused_linear import KfLinear


def _make_ln(hidden, fused):
    if fused:
        from kungfu_amd.ops.fused_ln import FusedLayerNorm

        return FusedLayerNorm(hidden)
    return nn.LayerNorm(hidden)


class BertLayer(nn.Module):
    def __init__(self, hidden, heads, ffn, fused_ln=False):
        super().__init__()
        self.heads = heads
        # KfLinear = nn.Linear + fused column-sum bias-grad kernel
        # (native autograd fallback outside the bf16-master conditions)
        self.qkv = KfLinear(hidden, 3 * hidden)
        self.proj = KfLinear(hidden, hidden)
        self.ln1 = _make_ln(hidden, fused_ln)
        self.fc1 = KfLinear(hidden, ffn)
        self.fc2 = KfLinear(ffn, hidden)
        self.ln2 = _make_ln(hidden, fused_ln)

    def forward(self, x):
        b, s, h = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, h // self.heads)
        q, k, v = (t.transpose(1, 2) for t in qkv.unbind(2))
        a = F.scaled_dot_product_attention(q, k, v)
        a = a.transpose(1, 2).reshape(b, s, h)
        x = self.ln1(x + self.proj(a))
        x = self.ln2(x + self.fc2(F.gelu(self.fc1(x))))
        return x


class Bert(nn.Module):
    def __init__(self, vocab=30522, hidden=768, layers=12, heads=12,
                 ffn=3072, max_len=512, fused_ln=False):
        super().__init__()
        self.fused_ln = fused_ln
        self.tok = nn.Embedding(vocab, hidden)
        self.pos = nn.Embedding(max_len, hidden)
        self.ln = _make_ln(hidden, fused_ln)
        self.blocks = nn.ModuleList(
            [BertLayer(hidden, heads, ffn, fused_ln=fused_ln)
             for _ in range(layers)])
        self.head = KfLinear(hidden, vocab)
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02)
                nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.02)

    def forward(self, ids):
        b, s = ids.shape
        pos = torch.arange(s, device=ids.device).unsqueeze(0)
        x = self.tok(ids) + self.pos(pos)
        if self.fused_ln and x.is_cuda:
            x = x.to(torch.bfloat16)  # embeddings stay fp32 under autocast
        x = self.ln(x)
        for blk in self.blocks:
            x = blk(x)
        return self.head(x)


def bert_base(vocab=30522, max_len=512, fused_ln=False):
    return Bert(vocab=vocab, max_len=max_len, fused_ln=fused_ln)
