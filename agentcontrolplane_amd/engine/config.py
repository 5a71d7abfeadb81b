"""Model + engine configuration.

Model presets are the architectures BASELINE.json names (Llama-3 8B/70B,
Mixtral 8x7B) plus tiny test configs; weights are random-init (no network),
shapes and dtypes are the real ones.
"""
from __future__ import annotations

import dataclasses
from typing import Optional


@dataclasses.dataclass
class ModelConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    # Qwen2-family attention bias on the QKV projection
    attention_bias: bool = False
    # MoE (Mixtral): 0 experts = dense
    num_experts: int = 0
    num_experts_per_tok: int = 2
    dtype: str = "bfloat16"

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    def params_bytes(self, dtype_bytes: int = 2) -> int:
        h, i, v, L = self.hidden_size, self.intermediate_size, self.vocab_size, self.num_layers
        kvh = self.num_kv_heads * self.head_dim
        qh = self.num_heads * self.head_dim
        attn = h * qh + 2 * h * kvh + qh * h
        mlp = 3 * h * i
        if self.is_moe:
            mlp = self.num_experts * 3 * h * i + h * self.num_experts
        per_layer = attn + mlp + 2 * h
        total = L * per_layer + 2 * v * h + h
        return total * dtype_bytes


PRESETS = {
    "llama3-8b": ModelConfig(),
    # Llama-2 family: same architecture, smaller rope base / context / vocab
    "llama2-7b": ModelConfig(
        name="llama2-7b", vocab_size=32000, hidden_size=4096,
        intermediate_size=11008, num_layers=32, num_heads=32,
        num_kv_heads=32, rope_theta=10000.0, rms_eps=1e-5, max_position=4096,
    ),
    "llama2-13b": ModelConfig(
        name="llama2-13b", vocab_size=32000, hidden_size=5120,
        intermediate_size=13824, num_layers=40, num_heads=40,
        num_kv_heads=40, rope_theta=10000.0, max_position=4096,
    ),
    # Qwen2: GQA + QKV bias (the one architectural delta from Llama)
    "qwen2-7b": ModelConfig(
        name="qwen2-7b", vocab_size=152064, hidden_size=3584,
        intermediate_size=18944, num_layers=28, num_heads=28,
        num_kv_heads=4, rope_theta=1000000.0, rms_eps=1e-6,
        max_position=8192, attention_bias=True,
    ),
    "llama3-70b": ModelConfig(
        name="llama3-70b",
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
    ),
    "mixtral-8x7b": ModelConfig(
        name="mixtral-8x7b",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        rope_theta=1000000.0,
        num_experts=8,
        num_experts_per_tok=2,
    ),
    # tiny GPU config: real head_dim/GQA shape for the CDNA4 kernels at
    # test scale (kernels require head_dim 128)
    "tiny-gpu": ModelConfig(
        name="tiny-gpu",
        vocab_size=512,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        max_position=2048,
        dtype="bfloat16",
    ),
    # tiny GPU config with Qwen2-style attention bias
    "tiny-gpu-bias": ModelConfig(
        name="tiny-gpu-bias",
        vocab_size=512,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        max_position=2048,
        attention_bias=True,
        dtype="bfloat16",
    ),
    # tiny configs for CPU tests
    # tiny bias variant (Qwen2-shaped attention for CPU tests)
    "tiny-bias": ModelConfig(
        name="tiny-bias",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        max_position=4096,
        attention_bias=True,
        dtype="float32",
    ),
    "tiny": ModelConfig(
        name="tiny",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        max_position=4096,
        dtype="float32",
    ),
    "tiny-moe": ModelConfig(
        name="tiny-moe",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        max_position=512,
        num_experts=4,
        num_experts_per_tok=2,
        dtype="float32",
    ),
}


@dataclasses.dataclass
class EngineConfig:
    model: str = "llama3-8b"
    device: str = "cuda"                 # "cuda" (ROCm) or "cpu"
    kv_block_size: int = 16
    # KV pool: fraction of free HBM to claim (288 GB/GPU on MI355X), or an
    # explicit block count for tests
    gpu_memory_utilization: float = 0.85
    num_kv_blocks: Optional[int] = None
    max_batch_size: int = 2048
    # chunked-prefill token budget per step: 16384 measured ~7% faster
    # wave p50 than 8192 at 1k concurrency (r02 A/B, gpurun_out/bench3b) —
    # fewer, larger steps amortize per-step host+launch cost; activations
    # at 16k tokens are ~1 GB against 288 GB HBM
    max_prefill_tokens: int = 16384
    max_model_len: int = 8192
    max_queue: int = 65536
    enforce_eager: bool = False          # True disables hipGraph capture
    # launch decode step t+1 from GPU-resident sampled tokens before step t
    # syncs (hides per-step host time); auto-disabled for TP and whenever a
    # grammar-constrained decode is running
    async_scheduling: bool = True
    # fold forced grammar byte runs (MID scaffolding, schema literals) into
    # one prefill chunk instead of sequential masked decode steps
    grammar_fold: bool = True
    tensor_parallel: int = 1
    seed: int = 0
    request_timeout_s: float = 600.0
    # HF-named safetensors dir; None = random-init (no network here)
    checkpoint_path: Optional[str] = None
    # HF tokenizer.json; defaults to <checkpoint_path>/tokenizer.json when
    # present, else the byte-level tokenizer (synthetic workloads)
    tokenizer_path: Optional[str] = None

    def model_config(self) -> ModelConfig:
        if self.model not in PRESETS:
            raise KeyError(f"unknown model preset {self.model!r}; have {sorted(PRESETS)}")
        return PRESETS[self.model]
