from rocps.obs.metrics import MetricsLogger, StepTimer, chrome_trace  # noqa: F401
from rocps.obs.events import EventFileWriter  # noqa: F401
