from petals_amd.models import register_block
from petals_amd.models.mixtral.block import MixtralBlock
from petals_amd.models.mixtral.config import MixtralConfig

register_block("mixtral")(MixtralBlock)


def _register_models():
    from petals_amd.models.mixtral.model import (
        DistributedMixtralForCausalLM,
        DistributedMixtralForSequenceClassification,
        DistributedMixtralModel,
    )
    from petals_amd.utils import auto_config

    auto_config.register_model_classes(
        "mixtral",
        config=MixtralConfig,
        model=DistributedMixtralModel,
        model_for_causal_lm=DistributedMixtralForCausalLM,
        model_for_sequence_classification=DistributedMixtralForSequenceClassification,
    )
