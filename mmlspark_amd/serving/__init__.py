from .server import (DistributedServingServer, LowLatencyGBDTScorer,  # noqa: F401
                     ServingServer)
