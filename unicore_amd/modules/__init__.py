"""isort:skip_file"""

from .layer_norm import LayerNorm
from .rms_norm import RMSNorm
from .softmax_dropout import softmax_dropout
from .gelu_dropout import gelu_dropout
from .dropout_add import dropout_add
from .gaussian import gaussian_basis
from .gated_mul import gated_mul
from .embedding import Embedding
from .multihead_attention import SelfMultiheadAttention, CrossMultiheadAttention
from .transformer_encoder_layer import TransformerEncoderLayer
from .transformer_encoder import (
    TransformerEncoder,
    init_bert_params,
    relative_position_bucket,
)
from .transformer_decoder_layer import TransformerDecoderLayer
from .transformer_decoder import TransformerDecoder
