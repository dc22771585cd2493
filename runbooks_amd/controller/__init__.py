"""Controllers (the operator).

Parity: reference internal/controller/ — one reconciler per kind
(Model/Dataset/Server/Notebook), a generic BuildReconciler instantiated per
kind, a ParamsReconciler, service-account/identity reconciliation, and the
manager wiring (reference cmd/controllermanager/main.go:40-241).
"""
from .build import BuildReconciler
from .dataset import DatasetReconciler
from .manager import ControllerManager, run_manager
from .model import ModelReconciler
from .notebook import NotebookReconciler
from .params import ParamsReconciler, mount_params_config_map, params_config_map_name
from .server import ServerReconciler
from .utils import Result, is_pod_ready, reconcile_job, resolve_env

__all__ = [
    "ModelReconciler", "DatasetReconciler", "ServerReconciler",
    "NotebookReconciler", "BuildReconciler", "ParamsReconciler",
    "ControllerManager", "run_manager", "Result",
    "reconcile_job", "is_pod_ready", "resolve_env",
    "mount_params_config_map", "params_config_map_name",
]
