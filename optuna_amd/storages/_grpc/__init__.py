from optuna_amd.storages._grpc.client import GrpcStorageProxy
from optuna_amd.storages._grpc.server import make_server, run_grpc_proxy_server


__all__ = ["GrpcStorageProxy", "make_server", "run_grpc_proxy_server"]
