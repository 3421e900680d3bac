from ant_ray_amd.models.llama import LlamaConfig, LlamaForCausalLM  # noqa: F401
from ant_ray_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel  # noqa: F401


def build_model(name: str, device="cuda", seq_len: int = 4096):
    name = name.lower()
    if name in ("llama3-8b", "llama-3-8b", "llama3_8b"):
        cfg = LlamaConfig.llama3_8b(max_seq=seq_len)
        return LlamaForCausalLM(cfg, device=device)
    if name in ("llama-tiny", "tiny"):
        cfg = LlamaConfig.tiny(max_seq=seq_len)
        return LlamaForCausalLM(cfg, device=device)
    if name in ("llama-tiny-d128", "tiny-d128"):
        # tiny model with the production head_dim (the decode/flash kernels
        # are D=128-only) — for GPU tests of the generation path
        cfg = LlamaConfig(hidden=512, n_layers=4, n_heads=4, n_kv_heads=2,
                          head_dim=128, intermediate=1024, vocab=1024,
                          max_seq=seq_len)
        return LlamaForCausalLM(cfg, device=device)
    if name in ("gpt2", "gpt2-small", "gpt2_small"):
        cfg = GPT2Config.small(max_seq=min(seq_len, 1024))
        return GPT2LMHeadModel(cfg, device=device)
    raise ValueError(f"unknown model {name}")
