"""Policies: selective activation checkpointing + strategy/dtype mapping.

Parity target: reference fms_fsdp/policies/ac_handler.py:16-64 (identical
evenly-spaced selection: 1-indexed block k is checkpointed when
k*p >= cut_off, cut_off starting at 1/2 and incrementing per hit — the
reference's unit test patterns transfer verbatim) and
fms_fsdp/policies/mixed_precision.py (dtype triples) /
train_utils.py:217-253 (strategy string mapping).

Instead of wrapping block modules in CheckpointWrapper (which would change
module identity under our runtime's hooks), we set a `_ac_enabled` flag the
block's forward consults, running its body under non-reentrant
torch.utils.checkpoint.
"""

import torch


def apply_selective_ac(model, block_class, p):
    """Enable AC on fraction p of `block_class` blocks, evenly spaced."""
    # fractions may arrive as strings like "1/3" from CLI (reference
    # ac_handler.py:47 uses eval; we parse the a/b form explicitly)
    if isinstance(p, str):
        if "/" in p:
            num, den = p.split("/", 1)
            p = float(num) / float(den)
        else:
            p = float(p)
    block_idx = 0
    cut_off = 1 / 2
    pattern = []
    for m in model.modules():
        if isinstance(m, block_class):
            block_idx += 1
            if block_idx * p >= cut_off:
                cut_off += 1
                m._ac_enabled = True
                pattern.append(True)
            else:
                m._ac_enabled = False
                pattern.append(False)
    return pattern


# (param_dtype, reduce_dtype) triples — reference mixed_precision.py:5-27
# (buffer dtype is moot here: our buffers are the fp32 rope tables, which
# the RoPE kernel reads in fp32 regardless)
_MP_POLICIES = {
    "fp16": (torch.float16, torch.float16),          # fpSixteen
    "bf16": (torch.bfloat16, torch.bfloat16),        # bfSixteen
    "bf16_working": (torch.float32, torch.bfloat16), # bfSixteen_working
    "fp32": (torch.float32, torch.float32),          # fp32_policy
}


def resolve_mp_policy(cfg):
    """cfg -> policy name (reference get_mixed_precision_policy,
    train_utils.py:192-214: bf16 when supported, else fp16 fallback)."""
    pol = (getattr(cfg, "mp_policy", "auto") or "auto").lower()
    if pol == "auto":
        if cfg.mixed_precision:
            pol = "bf16" if _bf16_ready() else "fp16"
        else:
            pol = "fp32"
    if pol not in _MP_POLICIES:
        raise ValueError(f"unknown mp_policy {pol} "
                         f"(one of auto|{'|'.join(_MP_POLICIES)})")
    return pol


def get_mixed_precision_dtypes(cfg):
    """cfg -> (param_dtype, reduce_dtype). cfg.reduce_dtype overrides the
    policy's reduce dtype when set to an explicit dtype name."""
    param_dtype, reduce_dtype = _MP_POLICIES[resolve_mp_policy(cfg)]
    override = (getattr(cfg, "reduce_dtype", "policy") or "policy").lower()
    names = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32}
    if override in names:
        reduce_dtype = names[override]
    return param_dtype, reduce_dtype


def needs_loss_scaling(cfg):
    """fp16 params can under/overflow without scaling; bf16/fp32 cannot."""
    return resolve_mp_policy(cfg) == "fp16"


def _bf16_ready():
    if not torch.cuda.is_available():
        return True  # CPU path: bf16 emulated, fine for tests
    return torch.cuda.is_bf16_supported()


def get_sharding_strategy(cfg):
    """cfg string -> strategy (reference train_utils.py:227-234)."""
    s = cfg.sharding_strategy.lower()
    if s not in ("fsdp", "hsdp", "ddp"):
        raise ValueError(f"sharding strategy {s} not supported (fsdp|hsdp|ddp)")
    return s


def resolve_reshard_after_forward(cfg, model_param_count=None):
    """'auto': reshard only when the gathered bf16 params would crowd the
    288 GB HBM (>=30B params); explicit bool passes through."""
    r = cfg.reshard_after_forward
    if isinstance(r, bool):
        return r
    if isinstance(r, str) and r.lower() in ("true", "false"):
        return r.lower() == "true"
    if model_param_count is None:
        return False
    return model_param_count >= 30e9
