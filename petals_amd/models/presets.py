"""Built-in model configs (no-network environment: benchmarks and tests build
models from these presets with random-init weights; real checkpoints load from
local safetensors dirs).

Shapes follow the public architecture specs of each model family.
"""

PRESETS = {
    # ------------------------------------------------------------- Llama
    "llama-2-7b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=32, intermediate_size=11008, vocab_size=32000,
        max_position_embeddings=4096, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "llama-2-13b": dict(
        model_type="llama", hidden_size=5120, num_hidden_layers=40, num_attention_heads=40,
        num_key_value_heads=40, intermediate_size=13824, vocab_size=32000,
        max_position_embeddings=4096, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "llama-2-70b": dict(
        model_type="llama", hidden_size=8192, num_hidden_layers=80, num_attention_heads=64,
        num_key_value_heads=8, intermediate_size=28672, vocab_size=32000,
        max_position_embeddings=4096, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "llama-3-8b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=128256,
        max_position_embeddings=8192, rms_norm_eps=1e-5, rope_theta=500000.0,
    ),
    "llama-3.1-70b": dict(
        model_type="llama", hidden_size=8192, num_hidden_layers=80, num_attention_heads=64,
        num_key_value_heads=8, intermediate_size=28672, vocab_size=128256,
        max_position_embeddings=131072, rms_norm_eps=1e-5, rope_theta=500000.0,
        rope_scaling=dict(rope_type="llama3", factor=8.0, low_freq_factor=1.0,
                          high_freq_factor=4.0, original_max_position_embeddings=8192),
    ),
    "llama-3.1-405b": dict(
        model_type="llama", hidden_size=16384, num_hidden_layers=126, num_attention_heads=128,
        num_key_value_heads=8, intermediate_size=53248, vocab_size=128256,
        max_position_embeddings=131072, rms_norm_eps=1e-5, rope_theta=500000.0,
        rope_scaling=dict(rope_type="llama3", factor=8.0, low_freq_factor=1.0,
                          high_freq_factor=4.0, original_max_position_embeddings=8192),
    ),
    "tinyllama-1.1b": dict(
        model_type="llama", hidden_size=2048, num_hidden_layers=22, num_attention_heads=32,
        num_key_value_heads=4, intermediate_size=5632, vocab_size=32000,
        max_position_embeddings=2048, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    # ------------------------------------------------------------- BLOOM
    "bloom-560m": dict(
        model_type="bloom", hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        vocab_size=250880, max_position_embeddings=2048, layer_norm_epsilon=1e-5,
    ),
    "bloom-7b1": dict(
        model_type="bloom", hidden_size=4096, num_hidden_layers=30, num_attention_heads=32,
        vocab_size=250880, max_position_embeddings=2048, layer_norm_epsilon=1e-5,
    ),
    "bloom-176b": dict(
        model_type="bloom", hidden_size=14336, num_hidden_layers=70, num_attention_heads=112,
        vocab_size=250880, max_position_embeddings=2048, layer_norm_epsilon=1e-5,
    ),
    # ------------------------------------------------------------ Falcon
    "falcon-7b": dict(
        model_type="falcon", hidden_size=4544, num_hidden_layers=32, num_attention_heads=71,
        num_kv_heads=1, vocab_size=65024, max_position_embeddings=2048,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=False, bias=False,
    ),
    "falcon-40b": dict(
        model_type="falcon", hidden_size=8192, num_hidden_layers=60, num_attention_heads=128,
        num_kv_heads=8, vocab_size=65024, max_position_embeddings=2048,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=True, bias=False,
    ),
    "falcon-180b": dict(
        model_type="falcon", hidden_size=14848, num_hidden_layers=80, num_attention_heads=232,
        num_kv_heads=8, vocab_size=65024, max_position_embeddings=2048,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=True, bias=False,
    ),
    # ----------------------------------------------------------- Mixtral
    "mixtral-8x7b": dict(
        model_type="mixtral", hidden_size=4096, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, intermediate_size=14336, vocab_size=32000,
        max_position_embeddings=32768, rms_norm_eps=1e-5, rope_theta=1000000.0,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    "mixtral-8x22b": dict(
        model_type="mixtral", hidden_size=6144, num_hidden_layers=56, num_attention_heads=48,
        num_key_value_heads=8, intermediate_size=16384, vocab_size=32768,
        max_position_embeddings=65536, rms_norm_eps=1e-5, rope_theta=1000000.0,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    # ------------------------------------------------ tiny test configs
    "test-llama-hd128": dict(
        model_type="llama", hidden_size=512, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=1024, vocab_size=128,
        max_position_embeddings=256, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "test-llama": dict(
        model_type="llama", hidden_size=64, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=128, vocab_size=128,
        max_position_embeddings=256, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "test-bloom": dict(
        model_type="bloom", hidden_size=64, num_hidden_layers=4, num_attention_heads=4,
        vocab_size=128, max_position_embeddings=256, layer_norm_epsilon=1e-5,
    ),
    "test-bloom-hd64": dict(  # fused-path geometry (head_dim 64) for GPU tests
        model_type="bloom", hidden_size=256, num_hidden_layers=4, num_attention_heads=4,
        vocab_size=128, max_position_embeddings=256, layer_norm_epsilon=1e-5,
    ),
    "test-falcon": dict(
        model_type="falcon", hidden_size=64, num_hidden_layers=4, num_attention_heads=4,
        num_kv_heads=2, vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=True, bias=False,
    ),
    "test-falcon-hd64": dict(  # fused-path geometry (head_dim 64, gq 2) for GPU tests
        model_type="falcon", hidden_size=256, num_hidden_layers=4, num_attention_heads=4,
        num_kv_heads=2, vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=True, bias=False,
    ),
    "test-falcon-gq29": dict(  # falcon-180b head geometry (gq=29, head_dim 64) for GPU tests
        model_type="falcon", hidden_size=1856, num_hidden_layers=2, num_attention_heads=29,
        num_kv_heads=1, vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=True, bias=False,
    ),
    "test-falcon-mqa71": dict(  # falcon-7b geometry (old decoder, MQA gq=71) for GPU tests
        model_type="falcon", hidden_size=4544, num_hidden_layers=2, num_attention_heads=71,
        vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=False,
        multi_query=True, bias=False,
    ),
    "test-falcon-mqa": dict(  # PRIME head count (5): exercises uneven TP head sharding
        model_type="falcon", hidden_size=80, num_hidden_layers=2, num_attention_heads=5,
        vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=True, new_decoder_architecture=False,
        multi_query=True, bias=False,
    ),
    "test-falcon-classic": dict(  # rw-style: sequential residual, biased linears
        model_type="falcon", hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
        vocab_size=128, max_position_embeddings=256,
        layer_norm_epsilon=1e-5, parallel_attn=False, new_decoder_architecture=False,
        multi_query=False, bias=True,
    ),
    "test-mixtral": dict(
        model_type="mixtral", hidden_size=64, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, intermediate_size=128, vocab_size=128,
        max_position_embeddings=256, rms_norm_eps=1e-5, rope_theta=10000.0,
        num_local_experts=4, num_experts_per_tok=2,
    ),
}
