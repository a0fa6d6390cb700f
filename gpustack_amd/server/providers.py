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


PROVIDERS = {"mock": MockProvider, "command": CommandProvider}


def get_provider(name: str, config: dict | None = None):
    cls = PROVIDERS.get(name)
    if cls is None:
        raise ValueError(f"unknown provider {name!r} "
                         f"(have: {sorted(PROVIDERS)})")
    return cls(config or {})
