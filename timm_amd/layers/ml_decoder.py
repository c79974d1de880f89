"""ML-Decoder multi-label classification head (reference
`timm/layers/ml_decoder.py`; paper arxiv 2111.12933).

Non-learnable group queries cross-attend the spatial features through one
transformer-decoder layer, then a grouped FC expands each query to its slice
of the class logits. The grouped FC is batched with a single bmm here
(the reference loops per query group).
"""
from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor, nn


def add_ml_decoder_head(model):
    """Swap a model's pooled classifier for an MLDecoder head."""
    num_classes = model.num_classes
    num_features = model.num_features
    if hasattr(model, 'global_pool') and hasattr(model, 'fc'):  # ResNet-style
        model.global_pool = nn.Identity()
        del model.fc
        model.fc = MLDecoder(num_classes=num_classes, initial_num_features=num_features)
    elif hasattr(model, 'global_pool') and hasattr(model, 'classifier'):  # EfficientNet-style
        model.global_pool = nn.Identity()
        del model.classifier
        model.classifier = MLDecoder(num_classes=num_classes, initial_num_features=num_features)
    elif 'RegNet' in model._get_name() or 'TResNet' in model._get_name():
        del model.head
        model.head = MLDecoder(num_classes=num_classes, initial_num_features=num_features)
    else:
        raise RuntimeError('Model head layout is not aligned with ml-decoder')
    if hasattr(model, 'drop_rate'):
        model.drop_rate = 0  # decoder carries its own dropout
    return model


class TransformerDecoderLayerOptimal(nn.Module):
    """Decoder layer without self-attention (queries only cross-attend)."""

    def __init__(self, d_model, nhead=8, dim_feedforward=2048, dropout=0.1, activation='relu',
                 layer_norm_eps=1e-5) -> None:
        super().__init__()
        self.norm1 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.dropout = nn.Dropout(dropout)
        self.dropout1 = nn.Dropout(dropout)
        self.dropout2 = nn.Dropout(dropout)
        self.dropout3 = nn.Dropout(dropout)
        self.multihead_attn = nn.MultiheadAttention(d_model, nhead, dropout=dropout)
        self.linear1 = nn.Linear(d_model, dim_feedforward)
        self.linear2 = nn.Linear(dim_feedforward, d_model)
        self.norm2 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.norm3 = nn.LayerNorm(d_model, eps=layer_norm_eps)
        self.activation = F.relu if activation == 'relu' else F.gelu

    def __setstate__(self, state):
        state.setdefault('activation', F.relu)
        super().__setstate__(state)

    def forward(self, tgt: Tensor, memory: Tensor, tgt_mask: Optional[Tensor] = None,
                memory_mask: Optional[Tensor] = None,
                tgt_key_padding_mask: Optional[Tensor] = None,
                memory_key_padding_mask: Optional[Tensor] = None) -> Tensor:
        tgt = self.norm1(tgt + self.dropout1(tgt))
        tgt = self.norm2(tgt + self.dropout2(self.multihead_attn(tgt, memory, memory)[0]))
        ff = self.linear2(self.dropout(self.activation(self.linear1(tgt))))
        return self.norm3(tgt + self.dropout3(ff))


class _DecoderStack(nn.Module):
    def __init__(self, layers):
        super().__init__()
        self.layers = nn.ModuleList(layers)

    def forward(self, tgt, memory):
        for layer in self.layers:
            tgt = layer(tgt, memory)
        return tgt


class MLDecoder(nn.Module):
    def __init__(self, num_classes, num_of_groups=-1, decoder_embedding=768, initial_num_features=2048):
        super().__init__()
        embed_len_decoder = 100 if num_of_groups < 0 else num_of_groups
        embed_len_decoder = min(embed_len_decoder, num_classes)
        self.embed_len_decoder = embed_len_decoder

        decoder_embedding = 768 if decoder_embedding < 0 else decoder_embedding
        self.embed_standart = nn.Linear(initial_num_features, decoder_embedding)

        layer = TransformerDecoderLayerOptimal(
            d_model=decoder_embedding, dim_feedforward=2048, dropout=0.1)
        # plain module list instead of nn.TransformerDecoder (whose newer
        # torch versions probe layer.self_attn, which this layer omits)
        self.decoder = _DecoderStack([layer])

        # frozen queries (paper finds learned queries unnecessary)
        self.query_embed = nn.Embedding(embed_len_decoder, decoder_embedding)
        self.query_embed.requires_grad_(False)

        # grouped FC: each query produces duplicate_factor logits
        self.num_classes = num_classes
        self.duplicate_factor = int(num_classes / embed_len_decoder + 0.999)
        self.duplicate_pooling = nn.Parameter(
            torch.empty(embed_len_decoder, decoder_embedding, self.duplicate_factor))
        self.duplicate_pooling_bias = nn.Parameter(torch.zeros(num_classes))
        nn.init.xavier_normal_(self.duplicate_pooling)

    def forward(self, x):
        if x.ndim == 4:  # NCHW feature map -> token sequence
            spatial = x.flatten(2).transpose(1, 2)
        else:
            spatial = x
        memory = F.relu(self.embed_standart(spatial), inplace=True)

        bs = memory.shape[0]
        tgt = self.query_embed.weight.unsqueeze(1).expand(-1, bs, -1)
        h = self.decoder(tgt, memory.transpose(0, 1)).transpose(0, 1)  # [bs, Q, E]

        # grouped FC as one batched matmul: [Q, bs, E] x [Q, E, dup]
        out = torch.bmm(h.transpose(0, 1), self.duplicate_pooling).transpose(0, 1)
        logits = out.reshape(bs, -1)[:, :self.num_classes] + self.duplicate_pooling_bias
        return logits
