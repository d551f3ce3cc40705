from petals_amd.models import register_block
from petals_amd.models.falcon.block import FalconBlock
from petals_amd.models.falcon.config import FalconConfig

register_block("falcon")(FalconBlock)


def _register_models():
    from petals_amd.models.falcon.model import (
        DistributedFalconForCausalLM,
        DistributedFalconForSequenceClassification,
        DistributedFalconModel,
    )
    from petals_amd.utils import auto_config

    auto_config.register_model_classes(
        "falcon",
        config=FalconConfig,
        model=DistributedFalconModel,
        model_for_causal_lm=DistributedFalconForCausalLM,
        model_for_sequence_classification=DistributedFalconForSequenceClassification,
    )
