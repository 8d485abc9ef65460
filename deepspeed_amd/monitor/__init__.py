from .monitor import (CSVMonitor, Monitor, MonitorMaster,  # noqa: F401
                      TensorBoardMonitor, WandbMonitor)
