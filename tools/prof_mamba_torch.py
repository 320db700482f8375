import torch, sys
sys.path.insert(0, ".")
from fms_fsdp_amd.config import get_model_config
from fms_fsdp_amd.models.mamba import MambaLMHeadModel, MambaBlock, MambaConfig
from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW

cfg = MambaConfig.from_dict(get_model_config("mamba_2.8b"))
torch.manual_seed(0)
with torch.device("cuda:0"):
    m = MambaLMHeadModel(cfg)
    m.reset_parameters()
m = m.bfloat16()
sm = ShardedModel(m, MambaBlock, sharding_strategy="fsdp",
                  param_dtype=torch.bfloat16)
opt = ShardedAdamW(sm, lr=1e-4)
x = torch.randint(0, cfg.vocab_size, (2, 4096), device="cuda:0")
y = torch.randint(0, cfg.vocab_size, (2, 4096), device="cuda:0")

def step():
    opt.zero_grad()
    loss = sm(x, labels=y)
    loss.backward()
    sm.clip_grad_norm_(1.0)
    opt.step()

for _ in range(2):
    step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CUDA]) as prof:
    step()
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=28))
