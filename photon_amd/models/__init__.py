from .mpt import MPTCausalLM, MPTConfig, build_model

__all__ = ["MPTCausalLM", "MPTConfig", "build_model"]
