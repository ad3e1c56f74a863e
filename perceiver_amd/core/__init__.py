from perceiver_amd.core.adapter import (
    ClassificationOutputAdapter,
    InputAdapter,
    OutputAdapter,
    QueryProvider,
    RotarySupport,
    TiedTokenOutputAdapter,
    TokenInputAdapter,
    TokenInputAdapterWithRotarySupport,
    TrainableQueryProvider,
)
from perceiver_amd.core.config import (
    CausalSequenceModelConfig,
    ClassificationDecoderConfig,
    DecoderConfig,
    EncoderConfig,
    PerceiverARConfig,
    PerceiverIOConfig,
)
from perceiver_amd.core.modules import (
    CausalSequenceModel,
    CrossAttention,
    CrossAttentionLayer,
    KVCache,
    MLP,
    MultiHeadAttention,
    PerceiverAR,
    PerceiverDecoder,
    PerceiverEncoder,
    PerceiverIO,
    SelfAttention,
    SelfAttentionBlock,
    SelfAttentionLayer,
)
from perceiver_amd.core.position import (
    FourierPositionEncoding,
    FrequencyPositionEncoding,
    RotaryPositionEmbedding,
    positions,
)
from perceiver_amd.core.utils import ModuleOutput, Residual, freeze, init_parameters
