from petals_amd.models import register_block
from petals_amd.models.llama.block import LlamaBlock
from petals_amd.models.llama.config import LlamaConfig

register_block("llama")(LlamaBlock)

# distributed model classes are imported lazily by utils.auto_config to avoid
# pulling the whole client stack into server processes


def _register_models():
    from petals_amd.models.llama.model import (
        DistributedLlamaForCausalLM,
        DistributedLlamaForSequenceClassification,
        DistributedLlamaModel,
    )
    from petals_amd.utils import auto_config

    auto_config.register_model_classes(
        "llama",
        config=LlamaConfig,
        model=DistributedLlamaModel,
        model_for_causal_lm=DistributedLlamaForCausalLM,
        model_for_sequence_classification=DistributedLlamaForSequenceClassification,
    )
