"""Server reconciler: Service + Deployment for HTTP model servers.

Parity: reference internal/controller/server_controller.go —
Reconcile (:50-75), model-readiness gating (:210-246), Deployment
(:114-205: replicas=1, readiness GET / on the `http-serve` port, model RO
mount at /content/model, `model-server` SA), Service (:307-335: 8080 →
http-serve), status from ReadyReplicas (:280-299).
"""
from __future__ import annotations

from ..api import conditions as cond
from ..api.types import Model, Server
from ..cloud import Mount, MountBucketConfig
from ..k8s import KubeClient
from .. import resources as res
from .params import ParamsReconciler, mount_params_config_map
from .utils import (
    Result,
    SA_MODEL_SERVER,
    reconcile_service_account,
    resolve_env,
)

HTTP_SERVE_PORT_NAME = "http-serve"


def _server_selector(server: Server) -> dict:
    return {"role": "run", "server": server.name}


class ServerReconciler:
    kind = "Server"

    def __init__(self, kube: KubeClient, cloud, sci_client):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.params = ParamsReconciler(kube)

    def reconcile(self, server: Server) -> Result:
        if not server.get_image():
            return Result()
        self.params.reconcile_params_config_map(server)
        return self._reconcile_server(server)

    def _gate(self, server: Server, reason: str) -> Result:
        server.set_status_ready(False)
        server.set_condition(cond.CONDITION_SERVING, False, reason,
                             observed_generation=server.generation)
        self.kube.update_status(server.to_dict())
        return Result()

    def _reconcile_server(self, server: Server) -> Result:
        raw = self.kube.get("substratus.ai/v1", "Model", server.namespace,
                            server.model.name)
        if raw is None:
            return self._gate(server, cond.REASON_MODEL_NOT_FOUND)
        model = Model.from_dict(raw)
        if not model.ready:
            return self._gate(server, cond.REASON_MODEL_NOT_READY)

        reconcile_service_account(self.cloud, self.sci, self.kube,
                                  server.namespace, SA_MODEL_SERVER)

        self.kube.apply(self._service(server))
        self.kube.apply(self._deployment(server, model))

        deploy = self.kube.get("apps/v1", "Deployment", server.namespace,
                               f"{server.name}-server") or {}
        ready_replicas = int((deploy.get("status") or {})
                             .get("readyReplicas", 0) or 0)
        if ready_replicas == 0:
            server.set_status_ready(False)
            server.set_condition(cond.CONDITION_SERVING, False,
                                 cond.REASON_DEPLOYMENT_NOT_READY,
                                 observed_generation=server.generation)
        else:
            server.set_status_ready(True)
            server.set_condition(cond.CONDITION_SERVING, True,
                                 cond.REASON_DEPLOYMENT_READY,
                                 observed_generation=server.generation)
        self.kube.update_status(server.to_dict())
        return Result(success=True)

    def _owner_ref(self, server: Server) -> dict:
        return {"apiVersion": "substratus.ai/v1", "kind": "Server",
                "name": server.name,
                "uid": server.metadata.get("uid", ""),
                "controller": True, "blockOwnerDeletion": True}

    def _service(self, server: Server) -> dict:
        return {
            "apiVersion": "v1", "kind": "Service",
            "metadata": {"name": f"{server.name}-server",
                         "namespace": server.namespace,
                         "ownerReferences": [self._owner_ref(server)]},
            "spec": {
                "selector": _server_selector(server),
                "ports": [{"name": "http", "protocol": "TCP", "port": 8080,
                           "targetPort": HTTP_SERVE_PORT_NAME}],
            },
        }

    def _deployment(self, server: Server, model: Model) -> dict:
        container_name = "serve"
        pod_meta = {
            "labels": _server_selector(server),
            "annotations": {
                "kubectl.kubernetes.io/default-container": container_name},
        }
        pod_spec = {
            "serviceAccountName": SA_MODEL_SERVER,
            "containers": [{
                "name": container_name,
                "image": server.get_image(),
                "imagePullPolicy": "Always",
                "command": list(server.command),
                "env": resolve_env(server.env),
                "ports": [{"name": HTTP_SERVE_PORT_NAME,
                           "containerPort": 8080}],
                "readinessProbe": {"httpGet": {
                    "path": "/", "port": HTTP_SERVE_PORT_NAME}},
            }],
        }
        mount_params_config_map(pod_spec, server, container_name)
        self.cloud.mount_bucket(pod_meta, pod_spec, model, MountBucketConfig(
            name="model", container=container_name,
            mounts=[Mount("artifacts", "model")], read_only=True))
        res.apply(pod_meta, pod_spec, container_name, self.cloud.name,
                  server.resources)
        return {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": {"name": f"{server.name}-server",
                         "namespace": server.namespace,
                         "ownerReferences": [self._owner_ref(server)]},
            "spec": {
                "replicas": 1,
                "selector": {"matchLabels": {"server": server.name}},
                "template": {"metadata": pod_meta, "spec": pod_spec},
            },
        }
