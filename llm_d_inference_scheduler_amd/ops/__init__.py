from .dispatch import (  # noqa: F401
    flash_prefill, hip_ops, move_blocks, paged_attention, reshape_and_cache,
    rmsnorm, rope, silu_mul,
)
