from rocps.obs.metrics import MetricsLogger, StepTimer, chrome_trace  # noqa: F401
