"""rocps flagship benchmark — BASELINE.json's headline metric:
images/sec (whole node), ResNet-50 sync-SGD, bf16, synthetic data,
random-init weights, at 1..8 MI355X.

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...

Each rank = one GPU worker with its co-located PS shard (reduce-scatter
+ fused-optimizer + all-gather over RCCL/xGMI). Rank 0 prints ONE JSON
line with the whole-job aggregate.
"""

import argparse
import json
import os
import time

import torch  # noqa: E402

# MIOpen exhaustive find on the conv configs (cached after warmup) —
# the fallback immediate-mode kernels are ~25x slower on some wrw convs.
torch.backends.cudnn.benchmark = True


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=256,
                    help="per-GPU batch size")
    ap.add_argument("--model", type=str, default="resnet50",
                    choices=["resnet50", "wide_resnet101_2", "bert_base",
                             "mnist_mlp"])
    ap.add_argument("--mode", type=str, default="sync",
                    choices=["sync", "async"])
    ap.add_argument("--num-ps", type=int, default=2,
                    help="async mode: host-DRAM PS shards (configs[2])")
    ap.add_argument("--bucket-mb", type=float, default=25.0)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["bf16", "fp32"])
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--torch-profile", type=str, default=None,
                    help="write a torch.profiler summary of the timed steps")
    ap.add_argument("--trace", type=str, default=None,
                    help="write a Chrome trace JSON of the timed steps")
    ap.add_argument("--hip-graph", action="store_true",
                    help="capture the train step in a hipGraph and replay "
                         "(sync SGD/momentum only; removes launch gaps)")
    return ap.parse_args()


def build(args, device, dtype):
    """Returns (model, optimizer, make_batch) for the chosen config."""
    from rocps.train.optimizer import AdamOptimizer, MomentumOptimizer
    torch.manual_seed(1234)
    if args.model in ("resnet50", "wide_resnet101_2"):
        from rocps.models.resnet import resnet50, wide_resnet101_2
        m = resnet50() if args.model == "resnet50" else wide_resnet101_2()
        m = m.to(device)
        if dtype == torch.bfloat16:
            m = m.bfloat16()
        m = m.to(memory_format=torch.channels_last)
        opt = MomentumOptimizer(0.256, 0.875)

        def make_batch(i):
            g = torch.Generator(device="cpu").manual_seed(i)
            x = torch.randn(args.batch, 3, 224, 224, generator=g).to(dtype)
            y = torch.randint(0, 1000, (args.batch,), generator=g)
            return (x.to(device, non_blocking=True)
                    .to(memory_format=torch.channels_last),
                    y.to(device, non_blocking=True))

        return m, opt, make_batch
    if args.model == "bert_base":
        from rocps.models.bert import bert_base
        m = bert_base().to(device)
        if dtype == torch.bfloat16:
            m = m.bfloat16()
        opt = AdamOptimizer(1e-4)

        def make_batch(i):
            g = torch.Generator(device="cpu").manual_seed(i)
            ids = torch.randint(0, 30522, (args.batch, args.seq_len),
                                generator=g)
            labels = torch.randint(0, 30522, (args.batch, args.seq_len),
                                   generator=g)
            return ids.to(device), labels.to(device)

        return m, opt, make_batch
    # mnist_mlp (CPU-friendly sanity config)
    from rocps.models import mnist_mlp

    m = mnist_mlp(hidden=128).to(device)
    opt = MomentumOptimizer(0.05, 0.9)

    def make_batch(i):
        g = torch.Generator(device="cpu").manual_seed(i)
        x = torch.rand(args.batch, 784, generator=g)
        y = torch.randint(0, 10, (args.batch,), generator=g)
        return x.to(device), y.to(device)

    return m, opt, make_batch


def main():
    args = parse_args()
    from rocps.engine.comm import Comm
    from rocps.engine.sync_engine import SyncEngine

    Comm.tune_env()
    comm = Comm.from_env()
    device = (torch.device(args.device) if args.device else comm.device)
    use_cuda = device.type == "cuda"
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    if not use_cuda and args.dtype == "bf16" and args.model != "mnist_mlp":
        dtype = torch.float32  # CPU sanity runs in fp32

    model, opt, make_batch = build(args, device, dtype)
    ps_procs = []
    if args.mode == "async":
        # configs[2]: N GPU workers + --num-ps host-DRAM PS shards on this
        # node. PS tasks are spawned as subprocesses by rank 0; gradient
        # flow goes via shared-memory push/apply/pull; the torchrun group
        # is used only for the timing barriers.
        import subprocess
        import sys as _sys

        from rocps.cluster import ClusterSpec, ControlClient
        from rocps.engine.async_engine import AsyncWorkerEngine

        base = int(os.environ.get("MASTER_PORT", "29500"))
        spec = {
            "ps": [f"127.0.0.1:{base + 101 + i}" for i in range(args.num_ps)],
            "worker": [f"127.0.0.1:{base + 201 + r}"
                       for r in range(comm.world_size)],
        }
        cluster = ClusterSpec(spec)
        if comm.rank == 0:
            env = dict(os.environ, CUDA_VISIBLE_DEVICES="",
                       HIP_VISIBLE_DEVICES="")
            for i in range(args.num_ps):
                ps_procs.append(subprocess.Popen(
                    [_sys.executable, "-m", "rocps.cli.ps_server",
                     "--cluster-json", json.dumps(spec),
                     "--task-index", str(i)], env=env))
        from rocps.cluster import Server
        server = Server(cluster, "worker", comm.rank)
        control = ControlClient(cluster.master_address)
        engine = AsyncWorkerEngine(model, cluster, control, comm.rank,
                                   device)
        engine.initialize(opt)
    else:
        engine = SyncEngine(model, comm, bucket_cap_mb=args.bucket_mb)
        engine.broadcast_parameters(src=0)

    n_batches = 4
    batches = [make_batch(1000 * comm.rank + i) for i in range(n_batches)]

    def eager_step(i):
        x, y = batches[i % n_batches]
        engine.zero_grad()
        if hasattr(opt, "begin_step"):
            opt.begin_step()
        loss = model.loss(x, y)
        engine.step(loss, opt)
        return loss

    step = eager_step
    if args.hip_graph and use_cuda:
        # capture fwd+bwd+comm+fused-optimizer in one hipGraph; replay
        # per step with a device-side copy of the rotating batch.
        x_s = batches[0][0].clone()
        y_s = batches[0][1].clone()

        def body():
            engine.zero_grad()
            loss = model.loss(x_s, y_s)
            engine.step(loss, opt)
            return loss

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                loss_static = body()
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            loss_static = body()

        def graph_step(i):
            x, y = batches[i % n_batches]
            x_s.copy_(x)
            y_s.copy_(y)
            graph.replay()
            return loss_static

        step = graph_step

    for i in range(args.warmup):
        step(i)
    comm.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    prof = None
    if args.torch_profile or args.trace:
        prof = torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA])
        prof.__enter__()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    comm.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if prof is not None:
        prof.__exit__(None, None, None)
        if comm.rank == 0:
            if args.torch_profile:
                with open(args.torch_profile, "w") as f:
                    f.write(prof.key_averages().table(
                        sort_by="self_cuda_time_total" if use_cuda
                        else "self_cpu_time_total", row_limit=60))
            if args.trace:
                prof.export_chrome_trace(args.trace)

    # max over ranks (conservative whole-job time)
    t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
    if comm.is_distributed:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    n = comm.world_size
    global_batch = args.batch * n
    if args.model == "bert_base":
        value = global_batch * args.seq_len * args.steps / elapsed
        metric, unit = "tokens/sec", "tokens/s"
    else:
        value = global_batch * args.steps / elapsed
        metric, unit = "images/sec", "images/s"
    if comm.rank == 0:
        out = {
            "metric": metric,
            "value": round(value, 1),
            "unit": unit,
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference published no numbers
            "dtype": ("bf16" if next(model.parameters()).dtype
                      == torch.bfloat16 else "fp32"),
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.seq_len if args.model == "bert_base" else None,
                "parallelism": f"dp{n}",
                "mode": args.mode,
                "num_ps": args.num_ps if args.mode == "async" else 0,
            },
        }
        print(json.dumps(out), flush=True)
    if args.mode == "async" and comm.rank == 0:
        from rocps.cluster import shutdown_cluster
        shutdown_cluster(cluster)
        for p in ps_procs:
            try:
                p.wait(timeout=10)
            except Exception:
                p.terminate()
    if comm.is_distributed:
        comm.destroy()


if __name__ == "__main__":
    main()
