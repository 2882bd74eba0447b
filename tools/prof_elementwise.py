"""Attribute the small elementwise CUDA kernels in the ResNet-50 step
to their aten ops / python stacks (round-2: ~160 vectorized_elementwise
launches per step in the torch profile with no obvious owner)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch  # noqa: E402

torch.backends.cudnn.benchmark = True


def main():
    from rocps.engine.comm import Comm
    from rocps.engine.sync_engine import SyncEngine
    from rocps.models.resnet import resnet50
    from rocps.train.optimizer import MomentumOptimizer

    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    m = resnet50().to(dev).bfloat16().to(memory_format=torch.channels_last)
    eng = SyncEngine(m, Comm(device=dev))
    opt = MomentumOptimizer(0.1, 0.9)
    x = (torch.randn(64, 3, 224, 224, device=dev).bfloat16()
         .to(memory_format=torch.channels_last))
    y = torch.randint(0, 1000, (64,), device=dev)

    def step():
        eng.zero_grad()
        loss = m.loss(x, y)
        eng.step(loss, opt)

    for _ in range(6):
        step()
    torch.cuda.synchronize()
    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA],
            with_stack=True) as prof:
        for _ in range(2):
            step()
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_stack_n=7).table(
        sort_by="cuda_time_total", row_limit=40, max_src_column_width=110))


if __name__ == "__main__":
    main()
