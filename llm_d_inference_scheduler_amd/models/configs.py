"""Model architecture configs (random-init weights; no network for checkpoints)."""
from dataclasses import dataclass


@dataclass(frozen=True)
class ModelConfig:
    name: str
    vocab_size: int
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    # Qwen3-style per-head RMSNorm of Q and K before RoPE
    qk_norm: bool = False
    # multimodal (encode role): simple ViT-style tower when set
    vision_hidden: int = 0
    vision_layers: int = 0
    vision_patches: int = 0

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def kv_bytes_per_token(self, dtype_bytes: int = 2) -> int:
        return 2 * self.num_layers * self.kv_size * dtype_bytes


LLAMA_3_8B = ModelConfig(
    name="llama-3-8b", vocab_size=128256, hidden_size=4096,
    intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
    head_dim=128, rope_theta=500000.0)

# Qwen3-32B — the model the reference's regression harness drives
# (config/manifests/regression-testing/single-workload-regression.yaml:
# Qwen3-32B, input 1024 / output 1024). GQA 64q/8kv, head_dim 128,
# per-head QK-RMSNorm; ~64 GB bf16 weights fit one MI355X beside a
# >100 GB KV pool.
QWEN3_32B = ModelConfig(
    name="qwen3-32b", vocab_size=151936, hidden_size=5120,
    intermediate_size=25600, num_layers=64, num_heads=64, num_kv_heads=8,
    head_dim=128, rope_theta=1000000.0, rms_eps=1e-6, qk_norm=True,
    max_position=8192)

TINY_QWEN = ModelConfig(
    name="tiny-qwen", vocab_size=1024, hidden_size=256,
    intermediate_size=512, num_layers=2, num_heads=8, num_kv_heads=4,
    head_dim=32, rope_theta=10000.0, max_position=512, qk_norm=True)

# LLaVA-1.5-7B language tower (Vicuna-7B shape) + ViT-L/14-336 vision tower
LLAVA_1_5_7B_TEXT = ModelConfig(
    name="llava-1.5-7b", vocab_size=32064, hidden_size=4096,
    intermediate_size=11008, num_layers=32, num_heads=32, num_kv_heads=32,
    head_dim=128, rope_theta=10000.0,
    vision_hidden=1024, vision_layers=24, vision_patches=576)

TINY_LLAMA = ModelConfig(
    name="tiny-llama", vocab_size=1024, hidden_size=256,
    intermediate_size=512, num_layers=2, num_heads=8, num_kv_heads=4,
    head_dim=32, rope_theta=10000.0, max_position=512)

TINY_LLAVA = ModelConfig(
    name="tiny-llava", vocab_size=1024, hidden_size=256,
    intermediate_size=512, num_layers=2, num_heads=8, num_kv_heads=4,
    head_dim=32, rope_theta=10000.0, max_position=512,
    vision_hidden=64, vision_layers=2, vision_patches=8)
