"""BERT-base — BASELINE.json configs[3]: data-parallel sync training
with fused Adam on 8×MI355X (the reference exercises the same stack
through TF MatMul/BatchMatMul/Gather/LayerNorm ops).

MI355X-first: bf16 activations, fused LayerNorm HIP kernel
(rocps.ops.layer_norm), rocBLAS/hipBLASLt GEMMs via nn.Linear, SDPA
attention (flash path on ROCm), MLM loss through the fused
softmax-xent kernel (vocab 30522).
"""

from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from rocps.ops.layer_norm import LayerNorm

__all__ = ["BertModel", "bert_base"]


class BertSelfAttention(nn.Module):
    def __init__(self, hidden: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = hidden // heads
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.out = nn.Linear(hidden, hidden)

    def forward(self, x):
        B, S, H = x.shape
        qkv = self.qkv(x).reshape(B, S, 3, self.heads, self.head_dim)
        q, k, v = qkv.unbind(dim=2)
        q = q.transpose(1, 2)  # B, heads, S, hd
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        return self.out(o.transpose(1, 2).reshape(B, S, H))


class BertLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, intermediate: int):
        super().__init__()
        self.attn = BertSelfAttention(hidden, heads)
        self.ln1 = LayerNorm(hidden)
        self.fc1 = nn.Linear(hidden, intermediate)
        self.fc2 = nn.Linear(intermediate, hidden)
        self.ln2 = LayerNorm(hidden)

    def _mlp(self, x):
        # ROCPS_BERT_FUSED=1: fc1+bias+GELU as ONE fused MFMA kernel
        # (C10's epilogue-3 variant emitting the pre-activation for
        # backward reuse, with the one-pass gelu_bwd kernel).  Measured
        # round 2 (profiles/bench_bert_*): the rocBLAS + fused-aten
        # composition below is ~5% faster at b64 AND b256 (the MFMA
        # kernel ties rocBLAS on this K=768 shape, and the epilogue's
        # extra pre-activation write is pure cost), so the default
        # stays on the library path; the fused kernels remain tested
        # and dispatchable.
        import os
        if (x.is_cuda and x.dtype == torch.bfloat16
                and os.environ.get("ROCPS_BERT_FUSED", "0") == "1"):
            from rocps.ops.gemm import (linear_gelu_train, matmul_nt,
                                        supported)
            B, S, H = x.shape
            inter = self.fc1.out_features
            if supported(B * S, inter, H):
                if torch.is_grad_enabled():
                    h = linear_gelu_train(x.reshape(B * S, H),
                                          self.fc1.weight, self.fc1.bias)
                else:
                    h = matmul_nt(x.reshape(B * S, H), self.fc1.weight,
                                  self.fc1.bias, gelu=True)
                return self.fc2(h.reshape(B, S, inter))
        return self.fc2(F.gelu(self.fc1(x), approximate="tanh"))

    def forward(self, x):
        x = self.ln1(x + self.attn(x))
        return self.ln2(x + self._mlp(x))


class BertModel(nn.Module):
    def __init__(self, vocab: int = 30522, hidden: int = 768,
                 layers: int = 12, heads: int = 12, intermediate: int = 3072,
                 max_pos: int = 512):
        super().__init__()
        self.tok_emb = nn.Embedding(vocab, hidden)
        self.pos_emb = nn.Embedding(max_pos, hidden)
        self.emb_ln = LayerNorm(hidden)
        self.layers = nn.ModuleList(
            [BertLayer(hidden, heads, intermediate) for _ in range(layers)])
        self.mlm_dense = nn.Linear(hidden, hidden)
        self.mlm_ln = LayerNorm(hidden)
        self.mlm_out = nn.Linear(hidden, vocab)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.tok_emb(input_ids) + self.pos_emb(pos)[None]
        x = self.emb_ln(x)
        for layer in self.layers:
            x = layer(x)
        x = self.mlm_ln(F.gelu(self.mlm_dense(x), approximate="tanh"))
        return self.mlm_out(x)  # (B, S, vocab)

    def loss(self, input_ids: torch.Tensor, labels: torch.Tensor):
        from rocps.ops.softmax_xent import softmax_cross_entropy_with_logits
        logits = self(input_ids)
        V = logits.shape[-1]
        return softmax_cross_entropy_with_logits(
            logits.reshape(-1, V), labels.reshape(-1)).mean()


def bert_base(**kw) -> BertModel:
    return BertModel(**kw)
