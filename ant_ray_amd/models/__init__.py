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


def setup_tunableop(local_rank: int = 0):
    """Load the committed hipBLASLt TunableOp table (profiles/
    tunableop_gfx950.csv) for this process, or (re)tune with
    ANTRAY_TUNE=1 (rows appended for shapes not yet in the table —
    decode/serve GEMMs are M=1..16 skinny shapes the training sweep
    never sees). Mirrors bench.py's setup; safe no-op on CPU."""
    import os

    if os.environ.get("ANTRAY_TUNEOP") == "0":
        return
    try:
        import torch.cuda.tunable as tun
    except ImportError:
        return
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    table = os.path.join(here, "..", "profiles", "tunableop_gfx950.csv")
    table = os.path.abspath(table)
    tuning = os.environ.get("ANTRAY_TUNE") == "1"
    if not tuning and not os.path.exists(table):
        return
    tun.enable(True)
    out = os.environ.get("ANTRAY_TUNE_OUT",
                         f"/tmp/tunableop_rank{local_rank}.csv")
    tun.set_filename(out if tuning else table)
    if tuning:
        if os.path.exists(table):
            try:
                tun.read_file(table)  # only UNSEEN shapes get tuned
            except Exception:
                pass
        tun.tuning_enable(True)
        tun.set_max_tuning_duration(100)
    else:
        tun.tuning_enable(False)
        try:
            tun.read_file(table)
        except Exception:
            pass
