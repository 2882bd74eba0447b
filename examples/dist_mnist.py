"""Distributed MNIST-MLP training — the reference repo's main script,
rebuilt on rocps (SURVEY.md §2a "main training script").

Run exactly like the reference (one command per process):

    python examples/dist_mnist.py --ps_hosts=127.0.0.1:2222 \
        --worker_hosts=127.0.0.1:2223,127.0.0.1:2224 \
        --job_name=ps --task_index=0
    python examples/dist_mnist.py ... --job_name=worker --task_index=0
    python examples/dist_mnist.py ... --job_name=worker --task_index=1 --issync=1

Sync mode (--issync): exactly-N gradient aggregation per step (rocps
SyncEngine: bucketed reduce-scatter/all-reduce + barrier semantics of
tf.train.SyncReplicasOptimizer). Async mode: host-DRAM PS shards.
Data is synthetic (deterministic per global step) — no network access.
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

import rocps  # noqa: E402
from rocps import flags  # noqa: E402
from rocps.models import mnist_mlp, synthetic_mnist_batch  # noqa: E402

flags.DEFINE_string("ps_hosts", "", "comma-separated host:port list")
flags.DEFINE_string("worker_hosts", "", "comma-separated host:port list")
flags.DEFINE_string("job_name", "", "'ps' or 'worker'")
flags.DEFINE_integer("task_index", 0, "task index within the job")
flags.DEFINE_boolean("issync", True, "synchronous (exactly-N) aggregation")
flags.DEFINE_integer("replicas_to_aggregate", 0,
                     "sync mode: aggregate K of N workers per step "
                     "(0 = all; K<N = the reference's backup-worker mode)")
flags.DEFINE_integer("train_steps", 50, "number of global steps")
flags.DEFINE_integer("batch_size", 32, "per-worker batch size")
flags.DEFINE_float("learning_rate", 0.1, "SGD learning rate")
flags.DEFINE_float("momentum", 0.0, "momentum (0 = plain SGD)")
flags.DEFINE_string("checkpoint_dir", "", "chief checkpoint dir (optional)")
flags.DEFINE_integer("save_checkpoint_steps", 0, "ckpt every N steps (0=off)")
flags.DEFINE_integer("hidden", 32, "MLP hidden width")
flags.DEFINE_integer("seed", 42, "data/model seed")
flags.DEFINE_string("loss_log", "", "write per-step losses (JSON) here")
flags.DEFINE_integer("kill_after_step", 0,
                     "fault injection: crash this worker after N local steps")
flags.DEFINE_boolean("elastic", False,
                     "sync mode: reform the collective group from the "
                     "surviving workers on peer death and restore the "
                     "latest checkpoint (rocps.train.recovery)")
FLAGS = flags.FLAGS


def main(_):
    cluster = rocps.train.ClusterSpec({
        "ps": [h for h in FLAGS.ps_hosts.split(",") if h],
        "worker": [h for h in FLAGS.worker_hosts.split(",") if h],
    })
    server = rocps.train.Server(cluster, FLAGS.job_name, FLAGS.task_index)

    nw = cluster.num_tasks("worker")
    semisync = bool(FLAGS.issync and FLAGS.replicas_to_aggregate
                    and FLAGS.replicas_to_aggregate < nw)
    if FLAGS.job_name == "ps":
        if not FLAGS.issync or semisync:
            from rocps.engine.async_engine import AsyncPSShard
            AsyncPSShard.serve_on(server, cluster)
        server.join()
        return 0

    num_workers = cluster.num_tasks("worker")
    is_chief = FLAGS.task_index == 0
    torch.manual_seed(FLAGS.seed)
    model = mnist_mlp(hidden=FLAGS.hidden)
    opt = (rocps.train.MomentumOptimizer(FLAGS.learning_rate,
                                         FLAGS.momentum)
           if FLAGS.momentum
           else rocps.train.GradientDescentOptimizer(FLAGS.learning_rate))
    if FLAGS.issync:
        opt = rocps.train.SyncReplicasOptimizer(
            opt,
            replicas_to_aggregate=(FLAGS.replicas_to_aggregate
                                   or num_workers),
            total_num_replicas=num_workers)
    trainer = rocps.train.Trainer(
        model, opt, cluster=cluster, job_name=FLAGS.job_name,
        task_index=FLAGS.task_index,
        mode="sync" if FLAGS.issync else "async")

    hooks = [rocps.train.StopAtStepHook(last_step=FLAGS.train_steps),
             rocps.train.StepCounterHook(every_n_steps=20,
                                         batch_size=FLAGS.batch_size * num_workers)]
    losses = []
    global_batch = FLAGS.batch_size * num_workers
    with rocps.train.MonitoredTrainingSession(
            trainer, is_chief=is_chief, hooks=hooks,
            checkpoint_dir=FLAGS.checkpoint_dir or None,
            save_checkpoint_steps=(FLAGS.save_checkpoint_steps or None),
            save_checkpoint_secs=(None if FLAGS.save_checkpoint_steps else 600),
    ) as sess:
        recovery = None
        if FLAGS.elastic and FLAGS.issync and trainer.mode == "sync":
            from rocps.train.recovery import ElasticRecovery
            recovery = ElasticRecovery(
                trainer, FLAGS.checkpoint_dir or None,
                comm_timeout=30.0, dead_threshold=5.0)
        local_step = 0
        while not sess.should_stop():
            # deterministic synthetic batch for this global step; each
            # worker takes its slice (the reference feeds next_batch());
            # under elastic recovery the slice follows the CURRENT rank
            # (the surviving workers re-cover the batch space)
            g = torch.Generator().manual_seed(FLAGS.seed + trainer.global_step)
            x, y = synthetic_mnist_batch(global_batch, generator=g)
            lo = trainer.comm.rank * FLAGS.batch_size
            xb = x[lo:lo + FLAGS.batch_size].to(trainer.device)
            yb = y[lo:lo + FLAGS.batch_size].to(trainer.device)
            try:
                if recovery is not None:
                    recovery.check_peers()
                loss = sess.run(lambda: trainer.model.loss(xb, yb))
            except Exception as e:  # noqa: BLE001 — peer loss mid-step
                if recovery is None:
                    raise
                print(f"[elastic] step failed ({type(e).__name__}: "
                      f"{str(e)[:200]}); reforming group", flush=True)
                err = None
                for _ in range(3):
                    try:
                        recovery.recover()
                        err = None
                        break
                    except Exception as e2:  # noqa: BLE001
                        err = e2
                        print(f"[elastic] recover attempt failed: "
                              f"{str(e2)[:200]}", flush=True)
                if err is not None:
                    raise err
                print(f"[elastic] regrouped: rank {trainer.comm.rank}/"
                      f"{trainer.comm.world_size} at step "
                      f"{trainer.global_step}", flush=True)
                continue
            losses.append(float(loss))
            local_step += 1
            if FLAGS.kill_after_step and local_step >= FLAGS.kill_after_step:
                os._exit(17)  # simulated crash (no cleanup, like a real one)

    if FLAGS.loss_log:
        # "{task}" lets a shared launcher argument fan out to one log
        # per worker (rocps-launch passes identical args to every task)
        with open(FLAGS.loss_log.replace("{task}", str(FLAGS.task_index)),
                  "w") as f:
            json.dump(losses, f)
    # every worker reports done; the chief shuts the cluster down only
    # once all workers finished (PS must keep serving until then).
    import time as _time
    trainer.control.incr("workers_done")
    if is_chief:
        # wait for every LIVE worker: dead ones (heartbeat silence) are
        # excluded, so a crashed peer cannot wedge the shutdown
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
