"""Engine and model configuration.

Model presets are random-init shapes (no network for checkpoints —
BASELINE.json: synthetic data / random weights); llama-3-8b is the flagship
benchmark config, llama-3-70b the TP=4+ config, tiny the CPU test config.
The engine-arg surface mirrors what the reference passes to SGLang via
role YAML (reference examples/inference/pd-disagg-standalone.yaml:96-135:
model, tp-size, disaggregation-mode, mem-fraction).
"""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class ModelConfig:
    name: str = "llama-3-8b"
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 16384
    dtype: str = "bfloat16"
    qk_norm: bool = False      # Qwen3-style per-head RMSNorm on q/k

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def kv_bytes_per_token(self) -> int:
        return 2 * self.num_layers * self.num_kv_heads * self.head_dim * 2

    @staticmethod
    def preset(name: str) -> "ModelConfig":
        if name in ("llama-3-8b", "llama3-8b", "8b"):
            return ModelConfig()
        if name in ("llama-3-70b", "llama3-70b", "70b"):
            return ModelConfig(name="llama-3-70b", hidden_size=8192,
                               intermediate_size=28672, num_layers=80,
                               num_heads=64, num_kv_heads=8, head_dim=128)
        if name in ("qwen3-32b", "qwen3", "32b"):
            # the reference KEP's benchmark model:
            # Qwen3-32B shapes with per-head q/k RMSNorm
            return ModelConfig(name="qwen3-32b", hidden_size=5120,
                               intermediate_size=25600, num_layers=64,
                               num_heads=64, num_kv_heads=8, head_dim=128,
                               vocab_size=151936, rope_theta=1e6,
                               rms_eps=1e-6, max_position=40960,
                               qk_norm=True)
        if name in ("qwen3-8b",):
            return ModelConfig(name="qwen3-8b", hidden_size=4096,
                               intermediate_size=12288, num_layers=36,
                               num_heads=32, num_kv_heads=8, head_dim=128,
                               vocab_size=151936, rope_theta=1e6,
                               rms_eps=1e-6, max_position=40960,
                               qk_norm=True)
        if name == "tiny-qwen":   # qk-norm CPU test shape
            return ModelConfig(name="tiny-qwen", hidden_size=256,
                               intermediate_size=512, num_layers=2,
                               num_heads=2, num_kv_heads=1, head_dim=128,
                               vocab_size=512, max_position=2048,
                               qk_norm=True)
        if name == "tiny":
            return ModelConfig(name="tiny", hidden_size=256,
                               intermediate_size=512, num_layers=2,
                               num_heads=2, num_kv_heads=1, head_dim=128,
                               vocab_size=512, max_position=2048)
        if name == "tiny-tp":   # TP-divisible CPU test shape
            return ModelConfig(name="tiny-tp", hidden_size=512,
                               intermediate_size=1024, num_layers=2,
                               num_heads=4, num_kv_heads=2, head_dim=128,
                               vocab_size=512, max_position=2048)
        raise ValueError(f"unknown model preset {name!r}")


@dataclass
class EngineConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    device: str = "cuda"
    page_size: int = 16
    max_batch_size: int = 256
    max_seq_len: int = 8192
    max_prefill_tokens: int = 8192     # per-step prefill token budget
    # KV pool sizing: fraction of FREE HBM taken after weights are resident
    gpu_memory_utilization: float = 0.85
    kv_pool_tokens: int = 0            # explicit override (tests)
    tp_size: int = 1
    tp_rank: int = 0
    enable_prefix_cache: bool = True   # full-page KV reuse (Mooncake analog)
    enforce_eager: bool = False        # False: capture decode in hipGraphs
    seed: int = 0
