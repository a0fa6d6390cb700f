"""Pydantic API payloads + re-exported table models."""
from __future__ import annotations

from pydantic import BaseModel, Field

from .tables import (  # noqa: F401
    ApiKey, Cluster,
    Benchmark,
    GPUInstance, GPUInstanceState, GPUInstanceTemplate, SSHPublicKey,
    Model, Org,
    ModelFile,
    ModelInstance,
    ModelInstanceState,
    ModelProvider,
    ModelRoute,
    ModelUsage,
    PlacementStrategy,
    RegistrationToken,
    SourceEnum,
    SystemLoad,
    User,
    Worker, WorkerPool,
    WorkerState,
)


class UserCreate(BaseModel):
    username: str
    password: str
    is_admin: bool = False
    full_name: str = ""
    org_id: int | None = None


class UserPublic(BaseModel):
    id: int
    username: str
    is_admin: bool
    full_name: str = ""


class LoginRequest(BaseModel):
    username: str
    password: str


class ApiKeyCreate(BaseModel):
    name: str
    expires_in: float | None = None


class ModelCreate(BaseModel):
    name: str
    source: str = SourceEnum.PRESET.value
    model_ref: str = "llama-3-8b"
    cluster_id: int | None = None  # None = any cluster
    org_id: int | None = None      # None = visible to all users
    description: str = ""
    replicas: int = 1
    categories: list[str] = Field(default_factory=lambda: ["llm"])
    placement_strategy: str = PlacementStrategy.BINPACK.value
    worker_selector: dict = Field(default_factory=dict)
    gpu_selector: dict | None = None
    gpus_per_replica: int = 1
    backend_parameters: dict = Field(default_factory=dict)
    env: dict = Field(default_factory=dict)
    max_model_len: int | None = None
    gpu_memory_utilization: float = 0.9
    speculative_config: dict | None = None
    extended_kv_cache: dict | None = None
    distributed_inference_across_workers: bool = False
    restart_on_error: bool = True
    scaling_schedule: dict | None = None
    lora_list: list[str] | None = None
    lora_adapters: list[dict] | None = None  # [{name, path}] dynamic LoRA


class ModelUpdate(BaseModel):
    replicas: int | None = None
    description: str | None = None
    backend_parameters: dict | None = None
    gpu_memory_utilization: float | None = None
    max_model_len: int | None = None


class GPUDeviceInfo(BaseModel):
    uuid: str = ""
    name: str = "AMD Instinct MI355X"
    vendor: str = "AMD"
    index: int = 0
    device_index: int = 0
    device_chip_index: int = 0
    arch_family: str = "gfx950"
    compute_capability: str = "gfx950"
    driver_version: str = ""
    runtime_version: str = ""
    type: str = "rocm"
    core: dict = Field(default_factory=lambda: {"total": 256, "utilization_rate": 0.0})
    memory: dict = Field(
        default_factory=lambda: {
            "total": 288 * 1024**3,
            "used": 0,
            "allocated": 0,
            "is_unified_memory": False,
        }
    )
    temperature: float = 0.0
    power_usage: float = 0.0


class WorkerRegister(BaseModel):
    name: str
    hostname: str = ""
    ip: str = ""
    port: int = 10150
    metrics_port: int = 10152
    labels: dict = Field(default_factory=dict)
    status: dict = Field(default_factory=dict)
    system_reserved: dict = Field(default_factory=dict)
    token: str = ""
    proxy_mode: str = "direct"


class WorkerStatusUpdate(BaseModel):
    status: dict = Field(default_factory=dict)
    state_message: str = ""


class ModelInstanceUpdate(BaseModel):
    state: str | None = None
    state_message: str | None = None
    port: int | None = None
    pid: int | None = None
    restart_count: int | None = None


class ModelRouteCreate(BaseModel):
    name: str
    targets: list[dict] = Field(default_factory=list)


class BenchmarkCreate(BaseModel):
    name: str
    model_name: str
    mode: str = "concurrency"      # "concurrency" | "qps"
    value: float = 8
    sweep: list[float] | None = None   # multi-point profile (one run per value)
    sla: dict | None = None            # SLA threshold overrides (analysis.py)
    duration_s: float = 30.0
    isl: int = 128
    osl: int = 64


class OrgCreate(BaseModel):
    name: str
    description: str = ""


class ClusterCreate(BaseModel):
    name: str
    description: str = ""


class WorkerPoolCreate(BaseModel):
    name: str
    provider: str = "mock"
    instance_type: str = "mi355x-8gpu"
    replicas: int = 0
    provider_config: dict = Field(default_factory=dict)
    labels: dict = Field(default_factory=dict)


class WorkerPoolUpdate(BaseModel):
    replicas: int | None = None
    provider_config: dict | None = None
    labels: dict | None = None


class GPUInstanceCreate(BaseModel):
    name: str
    flavor: str = "mi355x-1gpu"
    image: str = "rocm/dev-ubuntu-24.04"
    ssh_public_key: str = ""
    ssh_key_name: str | None = None     # reference a stored SSHPublicKey
    template: str | None = None         # GPUInstanceTemplate to start from
    provider: str = "k8s"
    provider_config: dict = Field(default_factory=dict)
    volumes: list[dict] = Field(default_factory=list)
    labels: dict = Field(default_factory=dict)


class GPUInstanceTemplateCreate(BaseModel):
    name: str
    flavor: str = "mi355x-1gpu"
    image: str = "rocm/dev-ubuntu-24.04"
    volumes: list[dict] = Field(default_factory=list)
    labels: dict = Field(default_factory=dict)
    provider: str = "k8s"
    provider_config: dict = Field(default_factory=dict)


class SSHPublicKeyCreate(BaseModel):
    name: str
    public_key: str


class ModelProviderCreate(BaseModel):
    name: str
    base_url: str
    api_key: str = ""
    models: list[str] | None = None
    enabled: bool = True
