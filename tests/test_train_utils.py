import torch

from fms_fsdp_amd.config import train_config
from fms_fsdp_amd.utils.train import LambdaLR, get_profiler


def test_profiler_factory():
    cfg = train_config()
    cfg.use_profiler = False
    assert get_profiler(cfg, 0) is None
    cfg.use_profiler = True
    assert get_profiler(cfg, 1) is None       # rank0-only gate
    p = get_profiler(cfg, 0)
    assert p is not None


def test_lambda_lr_matches_torch():
    lin = torch.nn.Linear(2, 2)
    topt = torch.optim.SGD(lin.parameters(), lr=0.1)
    tsched = torch.optim.lr_scheduler.LambdaLR(topt, lambda x: 1 / (x + 1))

    class FakeOpt:
        param_groups = [{"lr": 0.1}]
    f = FakeOpt()
    ours = LambdaLR(f, lambda x: 1 / (x + 1))
    for _ in range(5):
        assert abs(f.param_groups[0]["lr"] - topt.param_groups[0]["lr"]) < 1e-9
        tsched.step()
        ours.step()
