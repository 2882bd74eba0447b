"""Generic distributed training script — the reference's script shape
 over the full rocps model zoo.

    python -m rocps.cli.launch --num-ps 1 --num-workers 8 \
        examples/dist_train.py --model resnet50 --issync 1 \
        --batch_size 256 --train_steps 500

Models: mnist_mlp, resnet50, wide_resnet101_2, bert_base (synthetic
data, random-init weights). Sync mode = bucketed RCCL reduce-scatter /
all-gather; async = host-DRAM PS shards (use >=1 ps task).
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

import rocps  # noqa: E402
from rocps import flags  # noqa: E402
from rocps.obs import EventFileWriter, MetricsLogger  # noqa: E402

flags.DEFINE_string("ps_hosts", "", "comma-separated host:port list")
flags.DEFINE_string("worker_hosts", "", "comma-separated host:port list")
flags.DEFINE_string("job_name", "", "'ps' or 'worker'")
flags.DEFINE_integer("task_index", 0, "task index within the job")
flags.DEFINE_boolean("issync", True, "synchronous aggregation")
flags.DEFINE_string("model", "resnet50", "mnist_mlp|resnet50|"
                                         "wide_resnet101_2|bert_base")
flags.DEFINE_integer("train_steps", 100, "global steps")
flags.DEFINE_integer("batch_size", 64, "per-worker batch size")
flags.DEFINE_integer("seq_len", 128, "sequence length (bert)")
flags.DEFINE_float("learning_rate", 0.1, "base learning rate")
flags.DEFINE_float("momentum", 0.9, "momentum (conv nets)")
flags.DEFINE_string("checkpoint_dir", "", "chief checkpoint dir")
flags.DEFINE_integer("save_checkpoint_steps", 0, "ckpt every N steps")
flags.DEFINE_string("logdir", "", "metrics JSONL + TB events dir")
flags.DEFINE_string("dtype", "bf16", "bf16|fp32 (GPU compute dtype)")
flags.DEFINE_integer("seed", 42, "seed")
FLAGS = flags.FLAGS


def make_batch(model_name, batch, seq_len, step, seed, device):
    g = torch.Generator().manual_seed(seed + step)
    if model_name == "bert_base":
        ids = torch.randint(0, 30522, (batch, seq_len), generator=g)
        lbl = torch.randint(0, 30522, (batch, seq_len), generator=g)
        return ids.to(device), lbl.to(device)
    if model_name == "mnist_mlp":
        x = torch.rand(batch, 784, generator=g)
        y = torch.randint(0, 10, (batch,), generator=g)
        return x.to(device), y.to(device)
    x = torch.randn(batch, 3, 224, 224, generator=g)
    y = torch.randint(0, 1000, (batch,), generator=g)
    x = x.to(device)
    if device.type == "cuda":
        x = x.to(memory_format=torch.channels_last)
    return x, y.to(device)


def main(_):
    from rocps.models import build_model
    cluster = rocps.train.ClusterSpec({
        "ps": [h for h in FLAGS.ps_hosts.split(",") if h],
        "worker": [h for h in FLAGS.worker_hosts.split(",") if h],
    })
    server = rocps.train.Server(cluster, FLAGS.job_name, FLAGS.task_index)
    if FLAGS.job_name == "ps":
        if not FLAGS.issync:
            from rocps.engine.async_engine import AsyncPSShard
            AsyncPSShard.serve_on(server, cluster)
        server.join()
        return 0

    num_workers = cluster.num_tasks("worker")
    is_chief = FLAGS.task_index == 0
    torch.manual_seed(FLAGS.seed)
    torch.backends.cudnn.benchmark = True
    model = build_model(FLAGS.model)
    if torch.cuda.is_available():
        if FLAGS.dtype == "bf16":
            model = model.bfloat16()
        if FLAGS.model in ("resnet50", "wide_resnet101_2"):
            model = model.to(memory_format=torch.channels_last)
    if FLAGS.model == "bert_base":
        opt = rocps.train.AdamOptimizer(FLAGS.learning_rate)
    else:
        opt = rocps.train.MomentumOptimizer(FLAGS.learning_rate,
                                            FLAGS.momentum)
    if FLAGS.issync:
        opt = rocps.train.SyncReplicasOptimizer(
            opt, replicas_to_aggregate=num_workers,
            total_num_replicas=num_workers)
    trainer = rocps.train.Trainer(
        model, opt, cluster=cluster, job_name=FLAGS.job_name,
        task_index=FLAGS.task_index,
        mode="sync" if FLAGS.issync else "async")

    metrics = MetricsLogger(FLAGS.logdir or None, stdout_every=0)
    events = (EventFileWriter(FLAGS.logdir)
              if FLAGS.logdir and is_chief else None)
    hooks = [rocps.train.StopAtStepHook(last_step=FLAGS.train_steps),
             rocps.train.StepCounterHook(
                 every_n_steps=20, metrics=metrics,
                 batch_size=FLAGS.batch_size * num_workers)]
    with rocps.train.MonitoredTrainingSession(
            trainer, is_chief=is_chief, hooks=hooks,
            checkpoint_dir=FLAGS.checkpoint_dir or None,
            save_checkpoint_steps=(FLAGS.save_checkpoint_steps or None),
            save_checkpoint_secs=(None if FLAGS.save_checkpoint_steps
                                  else 600)) as sess:
        # input feeding (C14): producer threads synthesize the
        # step-keyed batches into a bounded pinned queue (the
        # reference's queue runners); the consumer's H2D copy is then
        # a pinned DMA that overlaps compute.  Seeds stay exact across
        # restores: producer index i feeds step base + i.
        base = trainer.global_step
        use_cuda = trainer.device.type == "cuda"
        qr = rocps.train.QueueRunner(
            lambda i: make_batch(FLAGS.model, FLAGS.batch_size,
                                 FLAGS.seq_len, base + i, FLAGS.seed,
                                 torch.device("cpu")),
            capacity=4, num_threads=2, pin_memory=use_cuda)
        qcoord = rocps.train.start_queue_runners([qr])
        try:
            while not sess.should_stop():
                x, y = qr.dequeue()
                x = x.to(trainer.device, non_blocking=True)
                y = y.to(trainer.device, non_blocking=True)
                loss = sess.run(lambda: trainer.model.loss(x, y))
                if events is not None and trainer.global_step % 10 == 0:
                    events.add_scalar("loss", float(loss),
                                      trainer.global_step)
        finally:
            qcoord.request_stop()
    if events is not None:
        events.close()

    import time as _time
    trainer.control.incr("workers_done")
    if is_chief:
        deadline = _time.time() + 300
        while _time.time() < deadline:
            done = trainer.control.read_counter("workers_done")
            dead = [m for m in (trainer.heartbeat.dead_members(threshold=6.0)
                                if trainer.heartbeat else [])
                    if m.startswith("worker:")]
            if done >= num_workers - len(dead):
                break
            _time.sleep(0.2)
        from rocps.cluster import shutdown_cluster
        shutdown_cluster(cluster)
    # tear the collective group down explicitly: gloo's background
    # threads racing interpreter exit abort with
    # "terminate called without an active exception"
    trainer.comm.destroy()
    return 0


if __name__ == "__main__":
    rocps.app.run(main)
