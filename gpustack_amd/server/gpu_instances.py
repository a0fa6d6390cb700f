"""On-demand SSH GPU instances — the gpustack-operator analog.

Reference: gpustack/gpu_instances/ (4.1k LoC) + the gpustack-operator Go
binary: users request an SSH-accessible GPU
container with an instance flavor (GPU count), image, public key and
volumes; an operator materializes it as a pod and reports the SSH
endpoint. Re-designed in-process for MI355X clusters: the server's
GPUInstanceController reconciles `gpu_instances` rows straight against
the Kubernetes API (utils/k8s_client.py) — pod with /dev/kfd + /dev/dri
device mounts and `amd.com/gpu` extended resources, plus a NodePort
service exposing sshd — no CRDs, no sidecar process, no webhook. A mock
provider serves tests and dry-runs (same split as worker pools,
server/providers.py).
"""
from __future__ import annotations

import logging
import queue
import shlex
import time
import uuid

from ..config import Config
from ..db import EventType, bus, get_session
from .controllers import _LeaderGated

logger = logging.getLogger(__name__)

# Instance flavors (reference: gpu_instances instance types / flavors).
# MI355X nodes partition naturally by GPU count; VRAM follows (288 GB per
# GPU of HBM3E).
FLAVORS: dict[str, dict] = {
    "mi355x-1gpu": {"gpus": 1, "cpu": "16", "memory": "128Gi"},
    "mi355x-2gpu": {"gpus": 2, "cpu": "32", "memory": "256Gi"},
    "mi355x-4gpu": {"gpus": 4, "cpu": "64", "memory": "512Gi"},
    "mi355x-8gpu": {"gpus": 8, "cpu": "128", "memory": "1Ti"},
}

_SSH_SETUP = (
    "mkdir -p /root/.ssh && echo {pubkey} > /root/.ssh/authorized_keys && "
    "chmod 700 /root/.ssh && chmod 600 /root/.ssh/authorized_keys && "
    "(which sshd >/dev/null 2>&1 || (apt-get update && "
    "apt-get install -y --no-install-recommends openssh-server)) && "
    "mkdir -p /run/sshd && exec /usr/sbin/sshd -D -e"
)


def instance_pod_manifest(inst: dict, namespace: str = "gpustack") -> dict:
    """Pod running sshd with the instance's public key, ROCm device nodes
    and the flavor's `amd.com/gpu` claim (AMD device plugin resource)."""
    flavor = FLAVORS.get(inst.get("flavor") or "", FLAVORS["mi355x-1gpu"])
    name = f"gpi-{inst['name']}"
    volumes = [{"name": "kfd", "hostPath": {"path": "/dev/kfd"}},
               {"name": "dri", "hostPath": {"path": "/dev/dri"}}]
    mounts = [{"name": "kfd", "mountPath": "/dev/kfd"},
              {"name": "dri", "mountPath": "/dev/dri"}]
    for i, vol in enumerate(inst.get("volumes") or []):
        vn = f"data-{i}"
        volumes.append({"name": vn, "emptyDir": {
            "sizeLimit": f"{int(vol.get('size_gb', 10))}Gi"}})
        mounts.append({"name": vn,
                       "mountPath": vol.get("mount_path", f"/data{i}")})
    return {
        "apiVersion": "v1", "kind": "Pod",
        "metadata": {
            "name": name, "namespace": namespace,
            "labels": {"app": "gpustack-amd-gpu-instance",
                       "gpustack.amd/instance": inst["name"],
                       **(inst.get("labels") or {})},
        },
        "spec": {
            "restartPolicy": "Never",
            "containers": [{
                "name": "instance",
                "image": inst.get("image") or "rocm/dev-ubuntu-24.04",
                "command": ["/bin/sh", "-c", _SSH_SETUP.format(
                    pubkey=shlex.quote(inst.get("ssh_public_key") or ""))],
                "ports": [{"containerPort": 22, "name": "ssh"}],
                "securityContext": {
                    "capabilities": {"add": ["SYS_PTRACE"]},
                    "seccompProfile": {"type": "Unconfined"},
                },
                "resources": {
                    "limits": {"amd.com/gpu": str(flavor["gpus"]),
                               "cpu": flavor["cpu"],
                               "memory": flavor["memory"]},
                },
                "volumeMounts": mounts,
            }],
            "volumes": volumes,
        },
    }


def instance_service_manifest(inst: dict, namespace: str = "gpustack") -> dict:
    """NodePort service exposing the instance's sshd."""
    return {
        "apiVersion": "v1", "kind": "Service",
        "metadata": {"name": f"gpi-{inst['name']}", "namespace": namespace},
        "spec": {
            "type": "NodePort",
            "selector": {"gpustack.amd/instance": inst["name"]},
            "ports": [{"name": "ssh", "port": 22, "targetPort": 22}],
        },
    }


class MockInstanceProvider:
    """In-process registry; `running_after` seconds simulates pod startup."""

    instances: dict[str, dict] = {}

    def __init__(self, config: dict | None = None):
        self.config = config or {}

    def create(self, inst: dict) -> str:
        eid = f"mock-{uuid.uuid4().hex[:8]}"
        MockInstanceProvider.instances[eid] = {
            "inst": dict(inst), "created_at": time.time(),
            "running_after": float(self.config.get("running_after", 0.0)),
        }
        return eid

    def status(self, external_id: str) -> dict:
        rec = MockInstanceProvider.instances.get(external_id)
        if rec is None:
            return {"phase": "gone"}
        if time.time() - rec["created_at"] >= rec["running_after"]:
            return {"phase": "running", "ssh_host": "mock.local",
                    "ssh_port": 2200}
        return {"phase": "creating"}

    def delete(self, external_id: str) -> None:
        MockInstanceProvider.instances.pop(external_id, None)


class K8sPodProvider:
    """Pod + NodePort service per instance through the minimal kube
    client; provider_config: api_server, token, namespace, verify."""

    def __init__(self, config: dict | None = None, client=None):
        cfg = config or {}
        if client is None:
            from ..utils.k8s_client import KubeClient

            client = KubeClient(api_server=cfg.get("api_server"),
                                token=cfg.get("token"),
                                namespace=cfg.get("namespace", "gpustack"),
                                verify=cfg.get("verify", True))
        self.kube = client
        self.namespace = self.kube.namespace

    def create(self, inst: dict) -> str:
        pod = instance_pod_manifest(inst, self.namespace)
        self.kube.create_pod(pod)
        self.kube.create_service(
            instance_service_manifest(inst, self.namespace))
        return pod["metadata"]["name"]

    def status(self, external_id: str) -> dict:
        pod = self.kube.get_pod(external_id)
        if pod is None:
            return {"phase": "gone"}
        phase = (pod.get("status") or {}).get("phase", "Pending")
        if phase != "Running":
            msg = (pod.get("status") or {}).get("reason", "")
            return {"phase": "failed" if phase == "Failed" else "creating",
                    "message": msg}
        host = (pod.get("status") or {}).get("hostIP", "")
        port = 0
        svc = self.kube.get_service(external_id)
        if svc is not None:
            for p in (svc.get("spec") or {}).get("ports", []):
                if p.get("name") == "ssh":
                    port = int(p.get("nodePort") or 0)
        return {"phase": "running", "ssh_host": host, "ssh_port": port}

    def delete(self, external_id: str) -> None:
        self.kube.delete_pod(external_id)
        self.kube.delete_service(external_id)


PROVIDERS = {"mock": MockInstanceProvider, "k8s": K8sPodProvider}


def get_provider(name: str, config: dict | None):
    cls = PROVIDERS.get(name)
    if cls is None:
        raise ValueError(f"unknown gpu-instance provider {name!r}")
    return cls(config or {})


class GPUInstanceController(_LeaderGated):
    """Reconciles gpu_instances rows against the provider (the operator
    loop, in-process): PENDING -> create -> CREATING -> poll until the
    pod runs -> RUNNING (ssh endpoint recorded); DELETING -> provider
    delete -> row removed; provider errors land in ERROR with message."""

    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._stop = False

    def stop(self) -> None:
        self._stop = True

    def run(self) -> None:
        q = bus.subscribe("gpu_instances")
        self.reconcile_all()
        while not self._stop:
            try:
                ev = q.get(timeout=10.0)
                if self._stop:
                    return
                if ev.type in (EventType.CREATED, EventType.UPDATED) \
                        and self._is_leader():
                    self.reconcile(ev.data["id"])
            except queue.Empty:
                if self._is_leader():
                    self.reconcile_all()

    def reconcile_all(self) -> None:
        from ..schemas import GPUInstance

        with get_session() as s:
            ids = [g.id for g in s.query(GPUInstance).all()]
        for gid in ids:
            try:
                self.reconcile(gid)
            except Exception:  # noqa: BLE001
                logger.exception("gpu instance %s reconcile failed", gid)

    def reconcile(self, gid: int) -> None:
        from ..db import ar_delete
        from ..schemas import GPUInstance, GPUInstanceState as St

        with get_session() as s:
            inst = s.get(GPUInstance, gid)
            if inst is None:
                return
            d = inst.to_dict()
        try:
            provider = get_provider(d["provider"], d.get("provider_config"))
        except Exception as e:  # noqa: BLE001
            self._set(gid, state=St.ERROR.value, state_message=str(e))
            return
        state = d["state"]
        try:
            if state == St.PENDING.value:
                eid = provider.create(d)
                self._set(gid, state=St.CREATING.value, external_id=eid,
                          state_message="")
            elif state in (St.CREATING.value, St.RUNNING.value):
                st = provider.status(d["external_id"])
                if st["phase"] == "running" and state != St.RUNNING.value:
                    self._set(gid, state=St.RUNNING.value,
                              ssh_host=st.get("ssh_host", ""),
                              ssh_port=int(st.get("ssh_port") or 0),
                              state_message="")
                elif st["phase"] in ("gone", "failed"):
                    self._set(gid, state=St.ERROR.value,
                              state_message=st.get("message")
                              or f"pod {st['phase']}")
            elif state == St.DELETING.value:
                if d["external_id"]:
                    provider.delete(d["external_id"])
                with get_session() as s:
                    inst = s.get(GPUInstance, gid)
                    if inst is not None:
                        ar_delete(s, inst)
        except Exception as e:  # noqa: BLE001
            logger.exception("gpu instance %s provider call failed", gid)
            self._set(gid, state=St.ERROR.value, state_message=str(e)[:500])

    @staticmethod
    def _set(gid: int, **fields) -> None:
        from ..db import ar_update
        from ..schemas import GPUInstance

        with get_session() as s:
            inst = s.get(GPUInstance, gid)
            if inst is None:
                return
            for k, v in fields.items():
                setattr(inst, k, v)
            ar_update(s, inst)
