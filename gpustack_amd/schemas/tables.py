"""SQLAlchemy tables — the data model (reference: gpustack/schemas/*).

Key parity points:
  * ModelInstance state machine (schemas/models.py:591-607):
    PENDING -> ANALYZING -> SCHEDULED -> INITIALIZING -> DOWNLOADING ->
    STARTING -> RUNNING (+ ERROR, UNREACHABLE)
  * Worker.status carries GPUDeviceInfo entries with type="rocm",
    arch_family="gfx950", 288 GiB HBM3E memory totals (schemas/workers.py:100)
  * Model carries replicas / placement / gpu_selector / backend params /
    speculative + extended-KV config (schemas/models.py:500)
"""
from __future__ import annotations

import enum
import time

from sqlalchemy import JSON, Boolean, Column, Float, ForeignKey, Integer, String, Text

from ..db import Base


class TimestampMixin:
    created_at = Column(Float, default=time.time, nullable=False)
    updated_at = Column(Float, default=time.time, onupdate=time.time, nullable=False)


class SerializeMixin:
    def to_dict(self) -> dict:
        return {c.name: getattr(self, c.name) for c in self.__table__.columns}

    def update_from(self, data: dict) -> None:
        for k, v in data.items():
            if hasattr(self, k) and k != "id":
                setattr(self, k, v)


class WorkerState(str, enum.Enum):
    NOT_READY = "not_ready"
    READY = "ready"
    UNREACHABLE = "unreachable"


class ModelInstanceState(str, enum.Enum):
    PENDING = "pending"
    ANALYZING = "analyzing"
    SCHEDULED = "scheduled"
    INITIALIZING = "initializing"
    DOWNLOADING = "downloading"
    STARTING = "starting"
    RUNNING = "running"
    ERROR = "error"
    UNREACHABLE = "unreachable"


class SourceEnum(str, enum.Enum):
    HUGGING_FACE = "huggingface"
    LOCAL_PATH = "local_path"
    PRESET = "preset"  # random-init named architecture (no-network serving)


class PlacementStrategy(str, enum.Enum):
    SPREAD = "spread"
    BINPACK = "binpack"


class Org(Base, TimestampMixin, SerializeMixin):
    """Tenant boundary (reference: schemas/users.py Org/Group multi-tenant
    RBAC — platform admin vs org members; models scoped per org)."""
    __tablename__ = "orgs"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    description = Column(Text, default="")


class User(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "users"
    id = Column(Integer, primary_key=True)
    username = Column(String(128), unique=True, nullable=False, index=True)
    hashed_password = Column(String(256), nullable=False)
    is_admin = Column(Boolean, default=False)
    full_name = Column(String(256), default="")
    require_password_change = Column(Boolean, default=False)
    org_id = Column(Integer, ForeignKey("orgs.id"), nullable=True, index=True)


class ApiKey(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "api_keys"
    id = Column(Integer, primary_key=True)
    user_id = Column(Integer, ForeignKey("users.id"), nullable=False)
    name = Column(String(128), nullable=False)
    access_key = Column(String(64), unique=True, index=True, nullable=False)
    hashed_secret = Column(String(256), nullable=False)
    expires_at = Column(Float, nullable=True)


class RegistrationToken(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "registration_tokens"
    id = Column(Integer, primary_key=True)
    token = Column(String(128), unique=True, index=True, nullable=False)
    description = Column(String(256), default="")
    cluster_id = Column(Integer, ForeignKey("clusters.id"), nullable=True)


class Cluster(Base, TimestampMixin, SerializeMixin):
    """Multi-cluster: workers register into a cluster via cluster-scoped
    registration tokens; models deploy into one cluster (reference:
    schemas/clusters.py — Docker/K8s/cloud clusters with per-cluster
    tokens and system principals)."""
    __tablename__ = "clusters"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    description = Column(Text, default="")
    is_default = Column(Boolean, default=False)


class Worker(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "workers"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False)
    cluster_id = Column(Integer, ForeignKey("clusters.id"), nullable=True,
                        index=True)
    hostname = Column(String(256), default="")
    ip = Column(String(64), default="")
    port = Column(Integer, default=10150)
    metrics_port = Column(Integer, default=10152)
    state = Column(String(32), default=WorkerState.NOT_READY.value)
    state_message = Column(Text, default="")
    labels = Column(JSON, default=dict)
    # WorkerStatus: cpu/memory/swap/filesystem/os/kernel/gpu_devices
    # (GPUDeviceInfo: uuid,name,vendor,index,arch_family,compute_capability,
    #  core{total,utilization_rate}, memory{total,used,allocated}, temperature,
    #  type="rocm")
    status = Column(JSON, default=dict)
    system_reserved = Column(JSON, default=dict)
    heartbeat_time = Column(Float, default=0.0)
    unreachable = Column(Boolean, default=False)
    proxy_mode = Column(String(16), default="direct")  # direct | tunnel


class Model(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "models"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    description = Column(Text, default="")
    source = Column(String(32), default=SourceEnum.PRESET.value)
    # PRESET: preset name (llama-3-8b...); LOCAL_PATH: dir with safetensors;
    # HUGGING_FACE: repo id (downloaded by the worker model-file manager)
    model_ref = Column(String(512), nullable=False)
    cluster_id = Column(Integer, ForeignKey("clusters.id"), nullable=True)
    # tenancy: None = visible to every user; set = that org's members only
    org_id = Column(Integer, ForeignKey("orgs.id"), nullable=True, index=True)
    replicas = Column(Integer, default=1)
    categories = Column(JSON, default=lambda: ["llm"])
    placement_strategy = Column(String(32), default=PlacementStrategy.BINPACK.value)
    worker_selector = Column(JSON, default=dict)     # label matching
    gpu_selector = Column(JSON, default=None)        # manual worker:gpu ids
    # device-class constraints (reference gpu_type_selector, vGPU slices /
    # MIG partitions, schemas/models.py:92-175; MI355X-native: AMD compute
    # partitioning — SPX/CPX — and NPS memory modes reported by amdsmi):
    # {"partition_compute": "CPX", "min_vram_gb": 32, "name_contains": "..."}
    gpu_type_selector = Column(JSON, default=None)
    gpus_per_replica = Column(Integer, default=1)    # TP degree per replica
    backend_parameters = Column(JSON, default=dict)  # engine kwargs overrides
    env = Column(JSON, default=dict)
    max_model_len = Column(Integer, default=None, nullable=True)
    gpu_memory_utilization = Column(Float, default=0.9)
    speculative_config = Column(JSON, default=None)  # {method: ngram/eagle3, ...}
    extended_kv_cache = Column(JSON, default=None)   # {ram_size/ram_ratio, ...}
    distributed_inference_across_workers = Column(Boolean, default=False)
    restart_on_error = Column(Boolean, default=True)
    lora_list = Column(JSON, default=None)  # adapter dirs merged at load
    # dynamic multi-LoRA: [{"name": ..., "path": ...}] served unmerged and
    # routable by adapter name (reference: per-LoRA child model routes,
    # gpustack/server/lora_model_routes.py)
    lora_adapters = Column(JSON, default=None)
    # cron-window autoscaling (reference: schemas/models.py:237-321):
    # {"rules": [{"cron": "0 9 * * 1-5", "duration_minutes": 60, "replicas": 4}]}
    scaling_schedule = Column(JSON, default=None)


class ModelInstance(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "model_instances"
    id = Column(Integer, primary_key=True)
    model_id = Column(Integer, ForeignKey("models.id"), nullable=False, index=True)
    model_name = Column(String(256), nullable=False)
    name = Column(String(256), unique=True, nullable=False)
    worker_id = Column(Integer, ForeignKey("workers.id"), nullable=True, index=True)
    worker_ip = Column(String(64), default="")
    gpu_indexes = Column(JSON, default=list)
    state = Column(String(32), default=ModelInstanceState.PENDING.value, index=True)
    state_message = Column(Text, default="")
    # {vram: {gpu_idx: bytes}, ram: bytes, kv_blocks: int} (schemas/models.py:623)
    computed_resource_claim = Column(JSON, default=dict)
    port = Column(Integer, nullable=True)
    pid = Column(Integer, nullable=True)
    restart_count = Column(Integer, default=0)
    distributed_servers = Column(JSON, default=None)  # subordinate workers
    # hash of the model's serving-relevant fields at creation; the
    # controller replaces instances whose hash no longer matches (model
    # updates redeploy, reference: model spec changes recreate instances)
    spec_hash = Column(String(64), default="")


class ModelFile(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "model_files"
    id = Column(Integer, primary_key=True)
    worker_id = Column(Integer, ForeignKey("workers.id"), nullable=False, index=True)
    source = Column(String(32), nullable=False)
    model_ref = Column(String(512), nullable=False)
    local_path = Column(String(1024), default="")
    state = Column(String(32), default="pending")   # pending/downloading/ready/error
    state_message = Column(Text, default="")
    size_bytes = Column(Integer, default=0)
    download_progress = Column(Float, default=0.0)


class ModelRoute(Base, TimestampMixin, SerializeMixin):
    """Decouples the published model name from deployments
    (reference: schemas/model_routes.py)."""
    __tablename__ = "model_routes"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    targets = Column(JSON, default=list)  # [{model_name, weight}]


class ModelProvider(Base, TimestampMixin, SerializeMixin):
    """External OpenAI-compatible providers routable through the gateway
    (reference: schemas/model_provider.py)."""
    __tablename__ = "model_providers"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False)
    base_url = Column(String(512), nullable=False)
    api_key = Column(String(512), default="")
    models = Column(JSON, default=None)  # None = any model name
    enabled = Column(Boolean, default=True)


class WorkerPool(Base, TimestampMixin, SerializeMixin):
    """Auto-provisioned worker capacity (reference: schemas/clusters.py
    WorkerPool + cloud_providers/; the provider abstraction is
    server/providers.py — mock / command-hook providers replace the
    reference's DigitalOcean droplets for bare-metal MI355X labs)."""
    __tablename__ = "worker_pools"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    provider = Column(String(64), default="mock")
    instance_type = Column(String(128), default="mi355x-8gpu")
    replicas = Column(Integer, default=0)
    provider_config = Column(JSON, default=dict)  # endpoint/commands/etc.
    labels = Column(JSON, default=dict)           # stamped on pool workers
    # provisioning records: [{instance_id, name, state, created_at}]
    instances = Column(JSON, default=list)
    state_message = Column(Text, default="")


class GPUInstanceState(str, enum.Enum):
    PENDING = "pending"
    CREATING = "creating"
    RUNNING = "running"
    DELETING = "deleting"
    ERROR = "error"


class GPUInstance(Base, TimestampMixin, SerializeMixin):
    """On-demand SSH-accessible GPU pod (reference: gpustack-operator CRDs
    + gpustack/gpu_instances/ — templates, flavors, persistent volumes).
    First-party equivalent: the server's GPUInstanceController drives the
    pod lifecycle directly through utils/k8s_client.py (or the mock
    provider for tests/dry-runs) — no Go sidecar process."""
    __tablename__ = "gpu_instances"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    flavor = Column(String(128), default="mi355x-1gpu")  # gpu count preset
    image = Column(String(512), default="rocm/dev-ubuntu-24.04")
    ssh_public_key = Column(Text, default="")
    provider = Column(String(64), default="k8s")         # k8s | mock
    provider_config = Column(JSON, default=dict)         # api_server/ns/...
    volumes = Column(JSON, default=list)  # [{name, size_gb, mount_path}]
    labels = Column(JSON, default=dict)
    state = Column(String(32), default=GPUInstanceState.PENDING.value)
    state_message = Column(Text, default="")
    external_id = Column(String(256), default="")        # pod name / mock id
    ssh_host = Column(String(256), default="")
    ssh_port = Column(Integer, default=0)


class GPUInstanceTemplate(Base, TimestampMixin, SerializeMixin):
    """Named GPU-instance preset (reference: gpu_instance_templates):
    flavor/image/volumes/labels captured once, instantiated many times."""
    __tablename__ = "gpu_instance_templates"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    flavor = Column(String(128), default="mi355x-1gpu")
    image = Column(String(512), default="rocm/dev-ubuntu-24.04")
    volumes = Column(JSON, default=list)
    labels = Column(JSON, default=dict)
    provider = Column(String(64), default="k8s")
    provider_config = Column(JSON, default=dict)


class SSHPublicKey(Base, TimestampMixin, SerializeMixin):
    """Reusable SSH public keys for GPU instances (reference:
    gpu_instance_ssh_public_keys)."""
    __tablename__ = "ssh_public_keys"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), unique=True, nullable=False, index=True)
    user_id = Column(Integer, index=True)
    public_key = Column(Text, nullable=False)


class Benchmark(Base, TimestampMixin, SerializeMixin):
    """In-product benchmark runs (reference: schemas/benchmark.py)."""
    __tablename__ = "benchmarks"
    id = Column(Integer, primary_key=True)
    name = Column(String(256), nullable=False)
    model_name = Column(String(256), nullable=False)
    worker_id = Column(Integer, nullable=True)
    state = Column(String(32), default="pending")  # pending/running/completed/error
    state_message = Column(Text, default="")
    # {mode: qps|concurrency, value, duration_s, isl, osl}
    config = Column(JSON, default=dict)
    results = Column(JSON, default=None)


class ModelUsage(Base, TimestampMixin, SerializeMixin):
    __tablename__ = "model_usage"
    id = Column(Integer, primary_key=True)
    user_id = Column(Integer, index=True)
    model_id = Column(Integer, index=True)
    model_name = Column(String(256), index=True)
    date = Column(String(16), index=True)  # YYYY-MM-DD
    prompt_tokens = Column(Integer, default=0)
    completion_tokens = Column(Integer, default=0)
    request_count = Column(Integer, default=0)


class ModelUsageArchive(Base, TimestampMixin, SerializeMixin):
    """Cold tier for usage rows (reference hot+archive table pairs,
    server/usage_archiver.py)."""
    __tablename__ = "model_usage_archive"
    id = Column(Integer, primary_key=True)
    user_id = Column(Integer, index=True)
    model_id = Column(Integer, index=True)
    model_name = Column(String(256), index=True)
    date = Column(String(16), index=True)
    prompt_tokens = Column(Integer, default=0)
    completion_tokens = Column(Integer, default=0)
    request_count = Column(Integer, default=0)


class ResourceEvent(Base, TimestampMixin, SerializeMixin):
    """Resource lifecycle events for metering (reference hot table of the
    resource-event pair, schemas/resource_events.py + the
    ResourceEventLogger wired at server/server.py:541-595): one row per
    instance state transition with the resource footprint attached, so
    billing/capacity pipelines can reconstruct GPU-seconds per model."""

    __tablename__ = "resource_events"
    id = Column(Integer, primary_key=True)
    event_type = Column(String(32), index=True)   # scheduled/running/stopped/error
    instance_id = Column(Integer, index=True)
    model_id = Column(Integer, index=True)
    model_name = Column(String(256), index=True)
    worker_id = Column(Integer, index=True)
    gpu_indexes = Column(JSON, default=list)
    vram_bytes = Column(Integer, default=0)       # total claim at event time
    ram_bytes = Column(Integer, default=0)
    timestamp = Column(Float, default=time.time, index=True)


class ResourceEventArchive(Base, TimestampMixin, SerializeMixin):
    """Cold tier of the resource-event pair."""

    __tablename__ = "resource_events_archive"
    id = Column(Integer, primary_key=True)
    event_type = Column(String(32), index=True)
    instance_id = Column(Integer, index=True)
    model_id = Column(Integer, index=True)
    model_name = Column(String(256), index=True)
    worker_id = Column(Integer, index=True)
    gpu_indexes = Column(JSON, default=list)
    vram_bytes = Column(Integer, default=0)
    ram_bytes = Column(Integer, default=0)
    timestamp = Column(Float, index=True)


class SystemLoad(Base, SerializeMixin):
    __tablename__ = "system_load"
    id = Column(Integer, primary_key=True)
    timestamp = Column(Float, default=time.time, index=True)
    cpu = Column(Float, default=0.0)
    ram = Column(Float, default=0.0)
    gpu = Column(Float, default=0.0)
    vram = Column(Float, default=0.0)
