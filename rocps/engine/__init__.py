from rocps.engine.comm import Comm  # noqa: F401
from rocps.engine.sync_engine import SyncEngine  # noqa: F401
