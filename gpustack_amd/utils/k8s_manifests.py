"""Kubernetes install manifests (reference: gpustack/k8s/ jinja templates —
manifest_template.py, daemonset.jinja — which deploy workers into a
cluster).

MI355X-native: plain dict->YAML generation (no jinja), with the ROCm
device plumbing a CDNA4 node needs — /dev/kfd + /dev/dri mounts, the
`amd.com/gpu` extended resource from the AMD device plugin, and the
video/render group supplement — instead of the reference's
vendor-matrixed runner images.
"""
from __future__ import annotations

import yaml

DEFAULT_IMAGE = "gpustack-amd:latest"


def server_manifests(namespace: str = "gpustack", image: str = DEFAULT_IMAGE,
                     bootstrap_password: str = "admin",
                     registration_token: str = "tok_cluster") -> list[dict]:
    labels = {"app": "gpustack-amd-server"}
    return [
        {"apiVersion": "v1", "kind": "Namespace",
         "metadata": {"name": namespace}},
        {"apiVersion": "v1", "kind": "Secret",
         "metadata": {"name": "gpustack-amd-bootstrap", "namespace": namespace},
         "stringData": {"bootstrap-password": bootstrap_password,
                        "registration-token": registration_token}},
        {"apiVersion": "apps/v1", "kind": "Deployment",
         "metadata": {"name": "gpustack-amd-server", "namespace": namespace},
         "spec": {
             "replicas": 1,
             "selector": {"matchLabels": labels},
             "template": {
                 "metadata": {"labels": labels},
                 "spec": {"containers": [{
                     "name": "server",
                     "image": image,
                     "command": ["python3", "-m", "gpustack_amd", "start"],
                     "args": ["--host", "0.0.0.0", "--port", "8080"],
                     "env": [
                         {"name": "GPUSTACK_BOOTSTRAP_PASSWORD",
                          "valueFrom": {"secretKeyRef": {
                              "name": "gpustack-amd-bootstrap",
                              "key": "bootstrap-password"}}},
                         {"name": "GPUSTACK_TOKEN",
                          "valueFrom": {"secretKeyRef": {
                              "name": "gpustack-amd-bootstrap",
                              "key": "registration-token"}}},
                     ],
                     "ports": [{"containerPort": 8080},
                               {"containerPort": 10151}],
                     "volumeMounts": [{"name": "data",
                                       "mountPath": "/root/.gpustack-amd"}],
                 }],
                     "volumes": [{"name": "data", "emptyDir": {}}]}}}},
        {"apiVersion": "v1", "kind": "Service",
         "metadata": {"name": "gpustack-amd-server", "namespace": namespace},
         "spec": {"selector": labels,
                  "ports": [{"name": "api", "port": 8080},
                            {"name": "metrics", "port": 10151}]}},
    ]


def worker_daemonset(namespace: str = "gpustack", image: str = DEFAULT_IMAGE,
                     server_url: str = "http://gpustack-amd-server:8080",
                     gpus_per_node: int = 8) -> dict:
    """ROCm worker DaemonSet (reference: k8s/daemonset.jinja). Requires the
    AMD GPU device plugin for the amd.com/gpu resource."""
    labels = {"app": "gpustack-amd-worker"}
    return {
        "apiVersion": "apps/v1", "kind": "DaemonSet",
        "metadata": {"name": "gpustack-amd-worker", "namespace": namespace},
        "spec": {
            "selector": {"matchLabels": labels},
            "template": {
                "metadata": {"labels": labels},
                "spec": {
                    "nodeSelector": {"gpustack.amd.com/worker": "true"},
                    "hostNetwork": True,
                    "containers": [{
                        "name": "worker",
                        "image": image,
                        "command": ["python3", "-m", "gpustack_amd", "start"],
                        "args": ["--server-url", server_url],
                        "env": [
                            {"name": "GPUSTACK_TOKEN",
                             "valueFrom": {"secretKeyRef": {
                                 "name": "gpustack-amd-bootstrap",
                                 "key": "registration-token"}}},
                            {"name": "GPUSTACK_WORKER_NAME",
                             "valueFrom": {"fieldRef": {
                                 "fieldPath": "spec.nodeName"}}},
                            # dmabuf IPC is required for RCCL / cross-process
                            # CUDA-tensor sharing on this driver stack
                            {"name": "HSA_ENABLE_IPC_MODE_LEGACY",
                             "value": "0"},
                        ],
                        "securityContext": {
                            "supplementalGroups": [44, 110],  # video, render
                        },
                        "resources": {"limits": {
                            "amd.com/gpu": gpus_per_node}},
                        "volumeMounts": [
                            {"name": "kfd", "mountPath": "/dev/kfd"},
                            {"name": "dri", "mountPath": "/dev/dri"},
                            {"name": "data",
                             "mountPath": "/root/.gpustack-amd"},
                        ],
                    }],
                    "volumes": [
                        {"name": "kfd", "hostPath": {"path": "/dev/kfd"}},
                        {"name": "dri", "hostPath": {"path": "/dev/dri"}},
                        {"name": "data",
                         "hostPath": {"path": "/var/lib/gpustack-amd",
                                      "type": "DirectoryOrCreate"}},
                    ],
                },
            },
        },
    }


def render_all(namespace: str = "gpustack", image: str = DEFAULT_IMAGE,
               server_url: str | None = None,
               bootstrap_password: str = "admin",
               registration_token: str = "tok_cluster",
               gpus_per_node: int = 8) -> str:
    docs = server_manifests(namespace, image, bootstrap_password,
                            registration_token)
    docs.append(worker_daemonset(
        namespace, image,
        server_url or f"http://gpustack-amd-server.{namespace}:8080",
        gpus_per_node))
    return yaml.safe_dump_all(docs, sort_keys=False)
