"""Backwards-compatibility aliases (parity: reference core/classifier.py)."""
from perceiver_amd.core.adapter import (  # noqa: F401
    ClassificationOutputAdapter,
    TrainableQueryProvider,
)
from perceiver_amd.core.config import ClassificationDecoderConfig  # noqa: F401
from perceiver_amd.core.modules import PerceiverDecoder as ClassificationDecoder  # noqa: F401
