"""Construction helpers shared by the task-model backends.

Every Perceiver IO task model pairs an encoder with a decoder built from the
same handful of pieces (a learned output-query array, an output adapter, a
``PerceiverDecoder`` shell). The reference inlines that assembly in each
backend file; here it is factored once so the task backends read as "encoder +
which head". The module tree (``encoder.*`` / ``decoder.*`` state-dict keys)
is unchanged — these helpers only centralize construction.
"""
from __future__ import annotations

from typing import Optional

from perceiver_amd.core import (
    ClassificationOutputAdapter,
    PerceiverDecoder,
    TrainableQueryProvider,
)


def latent_kwargs(config) -> dict:
    """Checkpointing/offload flags shared by encoder and decoder constructors."""
    return dict(
        activation_checkpointing=config.activation_checkpointing,
        activation_offloading=config.activation_offloading,
    )


def learned_queries(num_queries: int, num_channels: int, init_scale: float) -> TrainableQueryProvider:
    """A trainable output-query array (normal-init, scale ``init_scale``)."""
    return TrainableQueryProvider(
        num_queries=num_queries, num_query_channels=num_channels, init_scale=init_scale
    )


def assemble_decoder(adapter, queries, config, decoder_cfg) -> PerceiverDecoder:
    """Wrap an output adapter + query provider in the decoder cross-attention."""
    return PerceiverDecoder(
        output_adapter=adapter,
        output_query_provider=queries,
        num_latent_channels=config.num_latent_channels,
        **latent_kwargs(config),
        **decoder_cfg.base_kwargs(),
    )


def classification_decoder(config, num_queries: Optional[int] = None) -> PerceiverDecoder:
    """Classification head: learned query array + Linear-to-classes adapter.

    ``num_queries`` overrides the config's query count (the image classifier
    always decodes from a single query).
    """
    dc = config.decoder
    n = dc.num_output_queries if num_queries is None else num_queries
    return assemble_decoder(
        ClassificationOutputAdapter(
            num_classes=dc.num_classes,
            num_output_query_channels=dc.num_output_query_channels,
        ),
        learned_queries(n, dc.num_output_query_channels, dc.init_scale),
        config,
        dc,
    )
