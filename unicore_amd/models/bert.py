"""BERT model family (built-in).

Functional parity with the reference's example model
(reference examples/bert/model.py:18-260): learned positional embeddings +
relative-position-bucket bias encoder, BertLMHead with tied embedding
weights, optional classification heads; arches bert_base / bert_large / xlm.

CLI flags and architecture defaults are table-driven here (one dict per
arch) rather than a call per flag; the names/defaults match the reference
contract exactly.
"""

import logging

import torch
import torch.nn as nn
import torch.nn.functional as F

from unicore_amd import utils
from unicore_amd.models import (
    BaseUnicoreModel,
    register_model,
    register_model_architecture,
)
from unicore_amd.modules import LayerNorm, TransformerEncoder, init_bert_params
from unicore_amd.modules.embedding import Embedding

logger = logging.getLogger(__name__)

_ACTIVATIONS = ["relu", "gelu", "tanh", "linear"]

# flag -> argparse kwargs (contract: examples/bert/model.py:20-88)
_CLI_FLAGS = {
    "--encoder-layers": dict(type=int, metavar="L", help="num encoder layers"),
    "--encoder-embed-dim": dict(type=int, metavar="H",
                                help="encoder embedding dimension"),
    "--encoder-ffn-embed-dim": dict(type=int, metavar="F",
                                    help="encoder FFN hidden dimension"),
    "--encoder-attention-heads": dict(type=int, metavar="A",
                                      help="num encoder attention heads"),
    "--activation-fn": dict(choices=_ACTIVATIONS,
                            help="activation function to use"),
    "--pooler-activation-fn": dict(choices=_ACTIVATIONS,
                                   help="pooler-layer activation function"),
    "--emb-dropout": dict(type=float, metavar="D",
                          help="embedding dropout probability"),
    "--dropout": dict(type=float, metavar="D", help="dropout probability"),
    "--attention-dropout": dict(type=float, metavar="D",
                                help="attention-weight dropout probability"),
    "--activation-dropout": dict(type=float, metavar="D",
                                 help="FFN post-activation dropout"),
    "--pooler-dropout": dict(type=float, metavar="D",
                             help="masked-lm pooler dropout"),
    "--max-seq-len": dict(type=int,
                          help="number of positional embeddings to learn"),
    "--post-ln": dict(type=bool, help="post-LN (True) vs pre-LN (False)"),
}


@register_model("bert")
class BertModel(BaseUnicoreModel):
    @staticmethod
    def add_args(parser):
        for flag, spec in _CLI_FLAGS.items():
            parser.add_argument(flag, **spec)

    def __init__(self, args, dictionary):
        super().__init__()
        bert_default_arch(args)
        self.args = args
        self.padding_idx = dictionary.pad()
        dim = args.encoder_embed_dim
        self.embed_tokens = Embedding(len(dictionary), dim, self.padding_idx)
        self.embed_positions = Embedding(args.max_seq_len, dim)
        self.sentence_encoder = TransformerEncoder(
            encoder_layers=args.encoder_layers, embed_dim=dim,
            ffn_embed_dim=args.encoder_ffn_embed_dim,
            attention_heads=args.encoder_attention_heads,
            emb_dropout=args.emb_dropout, dropout=args.dropout,
            attention_dropout=args.attention_dropout,
            activation_dropout=args.activation_dropout,
            max_seq_len=args.max_seq_len, activation_fn=args.activation_fn,
            rel_pos=True, rel_pos_bins=32, max_rel_pos=128,
            post_ln=args.post_ln,
        )
        self.lm_head = BertLMHead(
            embed_dim=dim,
            output_dim=len(dictionary),
            activation_fn=args.activation_fn,
            weight=self.embed_tokens.weight,  # tied with input embedding
        )
        self.classification_heads = nn.ModuleDict()
        self.apply(init_bert_params)

    @classmethod
    def build_model(cls, args, task):
        return cls(args, task.dictionary)

    def forward(self, src_tokens, masked_tokens=None, features_only=False,
                classification_head_name=None, **kwargs):
        features_only = features_only or classification_head_name is not None

        pad_mask = src_tokens.eq(self.padding_idx)
        if pad_mask.is_cuda and torch.cuda.is_current_stream_capturing():
            pass  # .any() is a host sync — illegal inside hipGraph capture
        elif not pad_mask.any():
            pad_mask = None

        seq_len = src_tokens.size(1)
        out = self.embed_tokens(src_tokens)
        out = out + self.embed_positions.weight[:seq_len, :]
        out = self.sentence_encoder(out, padding_mask=pad_mask)

        if not features_only:
            out = self.lm_head(out, masked_tokens)
        if classification_head_name is not None:
            out = self.classification_heads[classification_head_name](out)
        return out

    def register_classification_head(self, name, num_classes=None,
                                     inner_dim=None, **kwargs):
        """Attach (or replace) a named sentence-classification head."""
        existing = (
            self.classification_heads[name]
            if name in self.classification_heads
            else None
        )
        if existing is not None:
            old_classes = existing.out_proj.out_features
            old_inner = existing.dense.out_features
            if (num_classes, inner_dim) != (old_classes, old_inner):
                logger.warning(
                    f're-registering head "{name}" with num_classes '
                    f"{num_classes} (prev: {old_classes}) and inner_dim "
                    f"{inner_dim} (prev: {old_inner})"
                )
        self.classification_heads[name] = BertClassificationHead(
            input_dim=self.args.encoder_embed_dim,
            inner_dim=inner_dim or self.args.encoder_embed_dim,
            num_classes=num_classes,
            activation_fn=self.args.pooler_activation_fn,
            pooler_dropout=self.args.pooler_dropout,
        )


class BertLMHead(nn.Module):
    """Masked-LM output head; the vocab projection shares the embedding
    matrix (plus its own bias)."""

    def __init__(self, embed_dim, output_dim, activation_fn, weight=None):
        super().__init__()
        self.dense = nn.Linear(embed_dim, embed_dim)
        self.act = utils.get_activation_fn(activation_fn)
        self.layer_norm = LayerNorm(embed_dim)
        if weight is None:
            weight = nn.Linear(embed_dim, output_dim, bias=False).weight
        self.weight = weight
        self.bias = nn.Parameter(torch.zeros(output_dim))

    def forward(self, features, masked_tokens=None, **kwargs):
        if masked_tokens is not None:
            # only the masked rows reach the vocab GEMM — big memory and
            # compute savings at 15% mask rate. A bool mask selects rows
            # directly; a flat long index (the pad-to-bucket path: row
            # count rounded up so the GEMM shape is batch-invariant and
            # tunable) gathers from the flattened tokens.
            if masked_tokens.dtype == torch.bool:
                features = features[masked_tokens, :]
            else:
                flat = features.reshape(-1, features.size(-1))
                features = flat.index_select(0, masked_tokens)
        h = self.layer_norm(self.act(self.dense(features)))
        return F.linear(h, self.weight) + self.bias


class BertClassificationHead(nn.Module):
    """[CLS]-pooled sentence classification head."""

    def __init__(self, input_dim, inner_dim, num_classes, activation_fn,
                 pooler_dropout):
        super().__init__()
        self.dense = nn.Linear(input_dim, inner_dim)
        self.act = utils.get_activation_fn(activation_fn)
        self.dropout = nn.Dropout(p=pooler_dropout)
        self.out_proj = nn.Linear(inner_dim, num_classes)

    def forward(self, features, **kwargs):
        pooled = self.dropout(features[:, 0, :])  # the [CLS] position
        pooled = self.dropout(self.act(self.dense(pooled)))
        return self.out_proj(pooled)


# architecture default tables (contract: examples/bert/model.py:223-260)

_BASE_DEFAULTS = dict(
    encoder_layers=12,
    encoder_embed_dim=768,
    encoder_ffn_embed_dim=3072,
    encoder_attention_heads=12,
    dropout=0.1,
    emb_dropout=0.1,
    attention_dropout=0.1,
    activation_dropout=0.0,
    pooler_dropout=0.0,
    max_seq_len=512,
    activation_fn="gelu",
    pooler_activation_fn="tanh",
    post_ln=True,
)


def _fill_defaults(args, table):
    for key, value in table.items():
        setattr(args, key, getattr(args, key, value))


@register_model_architecture("bert", "bert")
def bert_default_arch(args):
    _fill_defaults(args, _BASE_DEFAULTS)


# keep the reference's name importable for user code
base_architecture = bert_default_arch


@register_model_architecture("bert", "bert_base")
def bert_bert_default_arch(args):
    bert_default_arch(args)


@register_model_architecture("bert", "bert_large")
def bert_large_architecture(args):
    _fill_defaults(args, dict(
        encoder_layers=24,
        encoder_embed_dim=1024,
        encoder_ffn_embed_dim=4096,
        encoder_attention_heads=16,
    ))
    bert_default_arch(args)


@register_model_architecture("bert", "xlm")
def xlm_architecture(args):
    _fill_defaults(args, dict(
        encoder_layers=16,
        encoder_embed_dim=1280,
        encoder_ffn_embed_dim=1280 * 4,
        encoder_attention_heads=16,
    ))
    bert_default_arch(args)
