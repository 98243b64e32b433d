from .controller import ControllerLite  # noqa: F401
from .alerting import AlertEvaluator, AlertPolicy  # noqa: F401
from .cloud import CloudPoller, k8s_snapshot_to_platform  # noqa: F401
from .election import LeaderElector  # noqa: F401
from .grpc_server import GrpcSyncServer  # noqa: F401
