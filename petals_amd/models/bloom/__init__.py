from petals_amd.models import register_block
from petals_amd.models.bloom.block import BloomBlock
from petals_amd.models.bloom.config import BloomConfig

register_block("bloom")(BloomBlock)


def _register_models():
    from petals_amd.models.bloom.model import (
        DistributedBloomForCausalLM,
        DistributedBloomForSequenceClassification,
        DistributedBloomModel,
    )
    from petals_amd.utils import auto_config

    auto_config.register_model_classes(
        "bloom",
        config=BloomConfig,
        model=DistributedBloomModel,
        model_for_causal_lm=DistributedBloomForCausalLM,
        model_for_sequence_classification=DistributedBloomForSequenceClassification,
    )
