"""petals_amd — an MI355X-native swarm inference/fine-tuning engine for large LMs.

Capabilities modeled on bigscience-workshop/petals: transformer
blocks sharded by layer across independent servers, a thin client holding only
embeddings + LM head, DHT-based discovery, fault-tolerant autoregressive
inference with server-side KV caches, and parameter-efficient fine-tuning
(prompt tuning / LoRA) over the swarm.

The compute path is MI355X-first: hand-written HIP/CDNA4 kernels (MFMA + LDS
tiles) for the hot ops, hipBLASLt/rocBLAS for plain GEMMs, RCCL over xGMI for
intra-node activation hand-off and tensor parallelism.
"""

__version__ = "0.1.0"

from petals_amd.data_structures import (
    ModuleUID,
    RemoteModuleInfo,
    RemoteSpanInfo,
    ServerInfo,
    ServerState,
    parse_uid,
)

# Client API (imported lazily to keep `import petals_amd` light for servers)


def __getattr__(name):
    _client_names = {
        "AutoDistributedConfig",
        "AutoDistributedModel",
        "AutoDistributedModelForCausalLM",
        "AutoDistributedModelForSequenceClassification",
    }
    if name in _client_names:
        from petals_amd.utils import auto_config

        # ensure model families register themselves
        import petals_amd.models  # noqa: F401

        return getattr(auto_config, name)
    if name in ("RemoteSequential", "InferenceSession"):
        from petals_amd.client.remote_sequential import RemoteSequential
        from petals_amd.client.inference_session import InferenceSession

        return {"RemoteSequential": RemoteSequential, "InferenceSession": InferenceSession}[name]
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
