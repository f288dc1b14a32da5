"""Gang scheduling & topology: batch-scheduler plugins (volcano/yunikorn/
scheduler-plugins) and the MI355X-native xGMI gang scheduler."""
from .batchscheduler import (  # noqa: F401
    BatchScheduler,
    KaiBatchScheduler,
    SchedulerPluginsBatchScheduler,
    VolcanoBatchScheduler,
    XgmiGangScheduler,
    YunikornBatchScheduler,
    scheduler_for,
)
