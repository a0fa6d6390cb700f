"""Cloud / lab provisioning providers for worker pools.

Reference parity: gpustack/cloud_providers/ (abstract provider +
DigitalOcean droplets + cloud-init user data for worker pools). MI355X
clusters are usually bare-metal lab capacity, not droplets, so the
provider surface here is:

  * `mock`    — in-process registry; tests, dry-runs, capacity planning.
  * `command` — shell-hook provider: the pool's provider_config carries
    `create_command` / `delete_command` templates invoked with the
    instance name, type and bootstrap script in the environment. This is
    the integration point for lab tooling (ipmitool/redfish/slurm/
    internal CLIs) that actually powers MI355X nodes.

Every provider receives a BOOTSTRAP SCRIPT (the cloud-init analog) that
starts a worker agent registered against this server with the pool's
labels — reference: cloud_providers/user_data.py.
"""
from __future__ import annotations

import logging
import shlex
import subprocess
import time
import uuid

logger = logging.getLogger(__name__)


def bootstrap_script(server_url: str, registration_token: str,
                     labels: dict | None = None) -> str:
    """Worker bootstrap the provider runs on a fresh node (cloud-init
    analog, reference: cloud_providers/user_data.py)."""
    label_flags = " ".join(
        f"--label {shlex.quote(f'{k}={v}')}" for k, v in (labels or {}).items())
    return (
        "#!/bin/sh\n"
        f"exec python3 -m gpustack_amd start "
        f"--server-url {shlex.quote(server_url)} "
        f"--registration-token {shlex.quote(registration_token)} "
        f"{label_flags}\n"
    )


class MockProvider:
    """In-process instance registry (shared per process)."""

    instances: dict[str, dict] = {}

    def __init__(self, config: dict | None = None):
        self.config = config or {}

    def create(self, name: str, instance_type: str, user_data: str) -> str:
        iid = f"mock-{uuid.uuid4().hex[:8]}"
        MockProvider.instances[iid] = {
            "id": iid, "name": name, "type": instance_type,
            "user_data": user_data, "created_at": time.time(),
        }
        return iid

    def delete(self, instance_id: str) -> None:
        MockProvider.instances.pop(instance_id, None)

    def list(self) -> list[str]:
        return list(MockProvider.instances)


class CommandProvider:
    """Shell-hook provider: provider_config supplies
    `create_command` / `delete_command` templates. The create command gets
    GPUSTACK_INSTANCE_NAME / _TYPE / _USER_DATA in its environment and
    must print the instance id on stdout."""

    def __init__(self, config: dict):
        self.config = config
        if not config.get("create_command") or not config.get("delete_command"):
            raise ValueError("command provider needs create_command and "
                             "delete_command in provider_config")

    def create(self, name: str, instance_type: str, user_data: str) -> str:
        import os

        env = dict(os.environ,
                   GPUSTACK_INSTANCE_NAME=name,
                   GPUSTACK_INSTANCE_TYPE=instance_type,
                   GPUSTACK_USER_DATA=user_data)
        out = subprocess.run(self.config["create_command"], shell=True,
                             env=env, capture_output=True, text=True,
                             timeout=float(self.config.get("timeout", 300)))
        if out.returncode != 0:
            raise RuntimeError(f"create_command failed: {out.stderr.strip()}")
        iid = out.stdout.strip().splitlines()[-1] if out.stdout.strip() else name
        return iid

    def delete(self, instance_id: str) -> None:
        import os

        env = dict(os.environ, GPUSTACK_INSTANCE_ID=instance_id)
        out = subprocess.run(self.config["delete_command"], shell=True,
                             env=env, capture_output=True, text=True,
                             timeout=float(self.config.get("timeout", 300)))
        if out.returncode != 0:
            raise RuntimeError(f"delete_command failed: {out.stderr.strip()}")


class K8sProvider:
    """Worker pods in a Kubernetes cluster through the first-party kube
    client (utils/k8s_client.py — the same machinery the GPU-instance
    operator analog uses). Each pool replica is one ROCm worker pod
    (kfd/dri devices + `amd.com/gpu` claim) whose command runs the
    bootstrap script, registering against this server.

    provider_config: api_server, token, namespace, verify, image,
    gpus_per_worker (default 8)."""

    def __init__(self, config: dict, client=None):
        self.config = config or {}
        if client is None:
            from ..utils.k8s_client import KubeClient

            client = KubeClient(api_server=self.config.get("api_server"),
                                token=self.config.get("token"),
                                namespace=self.config.get("namespace",
                                                          "gpustack"),
                                verify=self.config.get("verify", True))
        self.kube = client

    def _manifest(self, name: str, instance_type: str, user_data: str) -> dict:
        gpus = int(self.config.get("gpus_per_worker", 8))
        return {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": name, "namespace": self.kube.namespace,
                         "labels": {"app": "gpustack-amd-pool-worker",
                                    "gpustack.amd/instance-type":
                                        instance_type}},
            "spec": {
                "restartPolicy": "Always",
                "containers": [{
                    "name": "worker",
                    "image": self.config.get("image", "gpustack-amd:latest"),
                    "command": ["/bin/sh", "-c", user_data],
                    "securityContext": {"capabilities": {
                        "add": ["SYS_PTRACE"]}},
                    "resources": {"limits": {"amd.com/gpu": str(gpus)}},
                    "volumeMounts": [
                        {"name": "kfd", "mountPath": "/dev/kfd"},
                        {"name": "dri", "mountPath": "/dev/dri"},
                    ],
                }],
                "volumes": [
                    {"name": "kfd", "hostPath": {"path": "/dev/kfd"}},
                    {"name": "dri", "hostPath": {"path": "/dev/dri"}},
                ],
            },
        }

    def create(self, name: str, instance_type: str, user_data: str) -> str:
        pod = self._manifest(name, instance_type, user_data)
        self.kube.create_pod(pod)
        return name

    def delete(self, instance_id: str) -> None:
        self.kube.delete_pod(instance_id)

    def list(self) -> list[str]:  # parity with MockProvider surface
        return []


PROVIDERS = {"mock": MockProvider, "command": CommandProvider,
             "k8s": K8sProvider}


def get_provider(name: str, config: dict | None = None):
    cls = PROVIDERS.get(name)
    if cls is None:
        raise ValueError(f"unknown provider {name!r} "
                         f"(have: {sorted(PROVIDERS)})")
    return cls(config or {})
