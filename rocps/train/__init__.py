"""``rocps.train`` — the reference's ``tf.train`` public surface.

Exports mirror the names the reference scripts use:
ClusterSpec, Server, replica_device_setter, Gradient/Momentum/Adam
optimizers, SyncReplicasOptimizer, MonitoredTrainingSession, Saver,
hooks, Coordinator, global-step helpers.
"""

from rocps.cluster import ClusterSpec, Server, ControlClient  # noqa: F401
from rocps.placement import replica_device_setter  # noqa: F401
from rocps.train.optimizer import (  # noqa: F401
    Optimizer, GradientDescentOptimizer, MomentumOptimizer, AdamOptimizer)
from rocps.train.sync_replicas import SyncReplicasOptimizer  # noqa: F401
from rocps.train.session import (  # noqa: F401
    MonitoredTrainingSession, Trainer)
from rocps.train.hooks import (  # noqa: F401
    SessionRunHook, StopAtStepHook, StepCounterHook, NanTensorHook,
    CheckpointSaverHook, SummarySaverHook, LoggingTensorHook,
    ProfilerHook)
from rocps.train.saver import (  # noqa: F401
    Saver, latest_checkpoint, NewCheckpointReader)
from rocps.train.coordinator import Coordinator, LooseVersionThread  # noqa: F401
from rocps.train.global_step import (  # noqa: F401
    get_or_create_global_step, get_global_step)
from rocps.train.queue_runner import (  # noqa: F401
    QueueRunner, start_queue_runners)
from rocps.train.recovery import ElasticRecovery, WorkerLost  # noqa: F401
from rocps.train.supervisor import Supervisor  # noqa: F401
from rocps.train.schedules import (  # noqa: F401
    exponential_decay, cosine_decay, polynomial_decay, LearningRateHook)
