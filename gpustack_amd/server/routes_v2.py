"""/v2 management API (reference: gpustack/routes/*).

CRUD for users/api-keys/workers/models/instances/routes plus the
`?watch=true` NDJSON streams that drive worker and controller sync
(reference watch mechanism: mixins/active_record.py:840,
client/generated_model_instance_client.py:165).
"""
from __future__ import annotations

import json
import time

from fastapi import APIRouter, Depends, HTTPException, Query, Request
from fastapi.responses import StreamingResponse

from ..db import Event, EventType, ar_create, ar_delete, ar_update, bus, get_session
from ..schemas import (
    ApiKey, ApiKeyCreate, Benchmark, BenchmarkCreate, ClusterCreate, Model,
    ModelCreate, OrgCreate,
    ModelInstance, ModelInstanceState, ModelInstanceUpdate, ModelProvider,
    ModelProviderCreate, ModelRoute, ModelRouteCreate, ModelUpdate,
    ModelUsage, RegistrationToken, SystemLoad, User, UserCreate, Worker,
    GPUInstanceCreate, GPUInstanceTemplateCreate, SSHPublicKeyCreate,
    WorkerPoolCreate, WorkerPoolUpdate, WorkerRegister, WorkerState,
    WorkerStatusUpdate,
)
from ..security import generate_api_key, generate_registration_token, hash_password
from .deps import get_admin_user, get_current_user, verify_worker_token

router = APIRouter(prefix="/v2")


def watch_ndjson(table: str, snapshot_rows: list[dict], flt):
    q = bus.subscribe(table)
    try:
        for row in snapshot_rows:
            yield json.dumps({"type": "CREATED", "data": row}) + "\n"
        import queue as _q

        while True:
            try:
                ev: Event = q.get(timeout=15.0)
            except _q.Empty:
                yield json.dumps({"type": "HEARTBEAT", "data": {}}) + "\n"
                continue
            if ev.type == EventType.HEARTBEAT or (flt and not flt(ev.data)):
                continue
            yield json.dumps({"type": ev.type.value, "data": ev.data}) + "\n"
    finally:
        bus.unsubscribe(table, q)


def _watch_stream(table: str, snapshot_rows: list[dict], flt):
    return StreamingResponse(watch_ndjson(table, snapshot_rows, flt),
                             media_type="application/x-ndjson")


# ---- users ---------------------------------------------------------------

@router.get("/users")
def list_users(_: User = Depends(get_admin_user)):
    with get_session() as s:
        return {"items": [u.to_dict() | {"hashed_password": None} for u in s.query(User).all()]}


@router.post("/users", status_code=201)
def create_user(body: UserCreate, _: User = Depends(get_admin_user)):
    with get_session() as s:
        if s.query(User).filter_by(username=body.username).first():
            raise HTTPException(409, "username exists")
        u = User(username=body.username, hashed_password=hash_password(body.password),
                 is_admin=body.is_admin, full_name=body.full_name,
                 org_id=body.org_id)
        ar_create(s, u)
        return u.to_dict() | {"hashed_password": None}


@router.delete("/users/{user_id}")
def delete_user(user_id: int, _: User = Depends(get_admin_user)):
    with get_session() as s:
        u = s.get(User, user_id)
        if not u:
            raise HTTPException(404)
        ar_delete(s, u)
        return {"ok": True}


# ---- api keys ------------------------------------------------------------

@router.get("/api_keys")
def list_api_keys(user: User = Depends(get_current_user)):
    with get_session() as s:
        keys = s.query(ApiKey).filter_by(user_id=user.id).all()
        return {"items": [k.to_dict() | {"hashed_secret": None} for k in keys]}


@router.post("/api_keys", status_code=201)
def create_api_key(body: ApiKeyCreate, user: User = Depends(get_current_user)):
    full, access, hashed = generate_api_key()
    with get_session() as s:
        k = ApiKey(user_id=user.id, name=body.name, access_key=access,
                   hashed_secret=hashed,
                   expires_at=(time.time() + body.expires_in) if body.expires_in else None)
        ar_create(s, k)
        return {"id": k.id, "name": k.name, "value": full}  # secret shown once


@router.delete("/api_keys/{key_id}")
def delete_api_key(key_id: int, user: User = Depends(get_current_user)):
    with get_session() as s:
        k = s.get(ApiKey, key_id)
        if not k or (k.user_id != user.id and not user.is_admin):
            raise HTTPException(404)
        ar_delete(s, k)
        return {"ok": True}


# ---- registration tokens -------------------------------------------------

@router.get("/tokens")
def list_tokens(_: User = Depends(get_admin_user)):
    with get_session() as s:
        return {"items": [t.to_dict() for t in s.query(RegistrationToken).all()]}


@router.post("/tokens", status_code=201)
def create_token(_: User = Depends(get_admin_user)):
    with get_session() as s:
        t = RegistrationToken(token=generate_registration_token())
        ar_create(s, t)
        return t.to_dict()


# ---- workers -------------------------------------------------------------

def _paginate(rows: list, page: int | None, per_page: int | None,
              search: str | None = None, search_key: str = "name") -> dict:
    """Reference-style list envelope: optional `search` substring filter on
    `search_key` plus `page`/`perPage` windowing with a pagination block.
    Without params the full item list returns unchanged (existing
    clients/watchers rely on that)."""
    if search:
        needle = search.lower()
        rows = [r for r in rows
                if needle in str(r.get(search_key, "")).lower()]
    total = len(rows)
    if page is None and per_page is None:
        return {"items": rows}
    per_page = max(1, per_page or 100)
    page = max(1, page or 1)
    lo = (page - 1) * per_page
    return {"items": rows[lo:lo + per_page],
            "pagination": {"page": page, "perPage": per_page,
                           "total": total,
                           "totalPage": (total + per_page - 1) // per_page}}


@router.get("/workers")
def list_workers(request: Request, watch: bool = Query(False),
                 page: int | None = Query(None),
                 perPage: int | None = Query(None),
                 search: str | None = Query(None),
                 user: User = Depends(get_current_user)):
    with get_session() as s:
        rows = [w.to_dict() for w in s.query(Worker).all()]
    if watch:
        return _watch_stream("workers", rows, None)
    return _paginate(rows, page, perPage, search)


def _token_cluster_id(request: Request) -> int | None:
    """Cluster the presented registration token is scoped to (multi-cluster:
    reference per-cluster registration tokens, schemas/clusters.py)."""
    auth = request.headers.get("authorization", "")
    token = auth.removeprefix("Bearer ").strip()
    if not token:
        return None
    with get_session() as s:
        row = s.query(RegistrationToken).filter_by(token=token).first()
        return row.cluster_id if row else None


@router.post("/workers/register")
def register_worker(body: WorkerRegister, request: Request,
                    _=Depends(verify_worker_token)):
    cluster_id = _token_cluster_id(request)
    with get_session() as s:
        w = s.query(Worker).filter_by(name=body.name).first()
        if w is None:
            w = Worker(name=body.name)
        if cluster_id is not None:
            w.cluster_id = cluster_id
        w.hostname = body.hostname
        w.ip = body.ip or (request.client.host if request.client else "")
        w.port = body.port
        w.metrics_port = body.metrics_port
        w.labels = body.labels
        w.status = body.status
        w.system_reserved = body.system_reserved
        w.proxy_mode = body.proxy_mode or "direct"
        w.heartbeat_time = time.time()
        w.state = WorkerState.READY.value
        if w.id is None:
            ar_create(s, w)
        else:
            ar_update(s, w)
        get_eval_cache().invalidate()  # placement landscape changed
        return w.to_dict()


class _WorkerStatusBuffer:
    """Batch worker status writes (reference:
    gpustack/server/worker_status_buffer.py): at fleet scale per-POST DB
    writes serialize on the session; posts land in a coalescing buffer
    (latest status per worker wins) that a flusher drains every interval
    in ONE transaction. State transitions (NOT_READY -> READY) still
    write through immediately so reconcilers see them without delay."""

    FLUSH_INTERVAL = 2.0

    def __init__(self):
        import threading

        self._lock = threading.Lock()
        self._pending: dict[int, dict] = {}
        self._timer: threading.Timer | None = None

    def put(self, worker_id: int, status: dict | None) -> None:
        import threading

        with self._lock:
            self._pending[worker_id] = {"status": status, "ts": time.time()}
            if self._timer is None:
                self._timer = threading.Timer(self.FLUSH_INTERVAL, self.flush)
                self._timer.daemon = True
                self._timer.start()

    def flush(self) -> None:
        with self._lock:
            batch, self._pending = self._pending, {}
            self._timer = None
        if not batch:
            return
        with get_session() as s:
            for wid, entry in batch.items():
                w = s.get(Worker, wid)
                if not w:
                    continue
                if entry["status"]:
                    w.status = entry["status"]
                w.heartbeat_time = entry["ts"]
            s.commit()


_status_buffer = _WorkerStatusBuffer()


@router.post("/workers/{worker_id}/status")
def worker_status(worker_id: int, body: WorkerStatusUpdate,
                  _=Depends(verify_worker_token)):
    with get_session() as s:
        w = s.get(Worker, worker_id)
        if not w:
            raise HTTPException(404, "worker not found (re-register)")
        if w.state != WorkerState.READY.value:
            # state transition: write through + publish for reconcilers
            w.status = body.status or w.status
            w.heartbeat_time = time.time()
            w.state = WorkerState.READY.value
            ar_update(s, w)
            return {"ok": True}
    _status_buffer.put(worker_id, body.status)
    return {"ok": True, "buffered": True}


@router.post("/workers/{worker_id}/heartbeat")
def worker_heartbeat(worker_id: int, _=Depends(verify_worker_token)):
    with get_session() as s:
        w = s.get(Worker, worker_id)
        if not w:
            raise HTTPException(404, "worker not found (re-register)")
        w.heartbeat_time = time.time()
        s.commit()
        return {"ok": True}


@router.delete("/workers/{worker_id}")
def delete_worker(worker_id: int, _: User = Depends(get_admin_user)):
    with get_session() as s:
        w = s.get(Worker, worker_id)
        if not w:
            raise HTTPException(404)
        ar_delete(s, w)
        get_eval_cache().invalidate()
        return {"ok": True}


# ---- models ----------------------------------------------------------------

@router.get("/models")
def list_models(watch: bool = Query(False),
                page: int | None = Query(None),
                perPage: int | None = Query(None),
                search: str | None = Query(None),
                categories: str | None = Query(None),
                user: User = Depends(get_current_user)):
    from .deps import model_allowed_for_user

    with get_session() as s:
        rows = [m.to_dict() for m in s.query(Model).all()
                if model_allowed_for_user(user, m)]
    if categories:
        want = set(categories.split(","))
        rows = [r for r in rows if want & set(r.get("categories") or [])]
    if watch:
        return _watch_stream("models", rows, None)
    return _paginate(rows, page, perPage, search)


@router.post("/models", status_code=201)
def create_model(body: ModelCreate, _: User = Depends(get_current_user)):
    with get_session() as s:
        if s.query(Model).filter_by(name=body.name).first():
            raise HTTPException(409, "model name exists")
        m = Model(**body.model_dump())
        if m.categories == ["llm"]:
            # auto-categorize from the architecture when the caller left
            # the default (reference: scheduler model_registry) — rerankers
            # and embedding checkpoints place and list correctly
            from ..utils.model_registry import categories_for_model

            cats = categories_for_model(m.source, m.model_ref)
            if cats:
                m.categories = cats
        ar_create(s, m)
        return m.to_dict()


@router.get("/models/{model_id}")
def get_model(model_id: int, _: User = Depends(get_current_user)):
    with get_session() as s:
        m = s.get(Model, model_id)
        if not m:
            raise HTTPException(404)
        return m.to_dict()


@router.patch("/models/{model_id}")
def update_model(model_id: int, body: ModelUpdate, _: User = Depends(get_current_user)):
    with get_session() as s:
        m = s.get(Model, model_id)
        if not m:
            raise HTTPException(404)
        m.update_from({k: v for k, v in body.model_dump().items() if v is not None})
        ar_update(s, m)
        return m.to_dict()


@router.delete("/models/{model_id}")
def delete_model(model_id: int, _: User = Depends(get_current_user)):
    with get_session() as s:
        m = s.get(Model, model_id)
        if not m:
            raise HTTPException(404)
        for inst in s.query(ModelInstance).filter_by(model_id=model_id).all():
            ar_delete(s, inst)
        ar_delete(s, m)
        return {"ok": True}


# ---- model instances -------------------------------------------------------

@router.get("/model_instances")
def list_instances(watch: bool = Query(False), worker_id: int | None = Query(None),
                   model_id: int | None = Query(None),
                   user: User = Depends(get_current_user)):
    with get_session() as s:
        q = s.query(ModelInstance)
        if model_id is not None:
            q = q.filter_by(model_id=model_id)
        rows = [i.to_dict() for i in q.all()]
    if worker_id is not None:
        # workers watch every instance event and filter locally on
        # worker_id so SCHEDULED-assignment events reach them; the snapshot
        # includes rows where this worker is a distributed subordinate
        def _mine(r):
            if r.get("worker_id") == worker_id:
                return True
            ds = r.get("distributed_servers") or {}
            return any(x.get("worker_id") == worker_id
                       for x in ds.get("subordinates", []))

        flt = None
        rows = [r for r in rows if _mine(r)]
    else:
        flt = None
    if watch:
        return _watch_stream("model_instances", rows, flt)
    return {"items": rows}


@router.patch("/model_instances/{instance_id}")
def update_instance(instance_id: int, body: ModelInstanceUpdate,
                    request: Request, _=Depends(verify_worker_token)):
    with get_session() as s:
        inst = s.get(ModelInstance, instance_id)
        if not inst:
            raise HTTPException(404)
        data = {k: v for k, v in body.model_dump().items() if v is not None}
        inst.update_from(data)
        ar_update(s, inst)
        return inst.to_dict()


@router.delete("/model_instances/{instance_id}")
def delete_instance(instance_id: int, _: User = Depends(get_current_user)):
    with get_session() as s:
        inst = s.get(ModelInstance, instance_id)
        if not inst:
            raise HTTPException(404)
        ar_delete(s, inst)
        return {"ok": True}


@router.post("/model_instances/{instance_id}/restart")
def restart_instance(instance_id: int, _: User = Depends(get_admin_user)):
    """Tear the instance down; the ModelController's replica reconcile
    recreates it (the reference's restart semantics — state machine runs
    PENDING→…→RUNNING again on a fresh process)."""
    with get_session() as s:
        inst = s.get(ModelInstance, instance_id)
        if not inst:
            raise HTTPException(404)
        model_id = inst.model_id
        ar_delete(s, inst)
    return {"status": "restarting", "model_id": model_id}


# ---- model routes ----------------------------------------------------------

@router.get("/model_files")
def list_model_files(_: User = Depends(get_current_user)):
    from ..schemas import ModelFile

    with get_session() as s:
        return {"items": [f.to_dict() for f in s.query(ModelFile).all()]}


@router.post("/model_files")
def upsert_model_file(body: dict, request: Request,
                      _=Depends(verify_worker_token)):
    """Worker-reported local model artifacts (reference: ModelFile records,
    schemas/model_files.py) — feeds the scheduler's locality scorer."""
    from ..schemas import ModelFile

    with get_session() as s:
        row = s.query(ModelFile).filter_by(
            worker_id=body.get("worker_id"), source=body.get("source"),
            model_ref=body.get("model_ref")).first()
        if row is None:
            row = ModelFile(worker_id=body["worker_id"],
                            source=body.get("source", "huggingface"),
                            model_ref=body["model_ref"])
            s.add(row)
        row.local_path = body.get("local_path", row.local_path)
        row.size_bytes = body.get("size_bytes", row.size_bytes)
        row.state = body.get("state", "ready")
        s.commit()
        return row.to_dict()


@router.get("/orgs")
def list_orgs(_: User = Depends(get_current_user)):
    from ..schemas import Org

    with get_session() as s:
        return {"items": [o.to_dict() for o in s.query(Org).all()]}


@router.post("/orgs", status_code=201)
def create_org(body: OrgCreate, _: User = Depends(get_admin_user)):
    from ..schemas import Org

    with get_session() as s:
        if s.query(Org).filter_by(name=body.name).first():
            raise HTTPException(409, "org exists")
        o = Org(name=body.name, description=body.description)
        ar_create(s, o)
        return o.to_dict()


@router.delete("/orgs/{org_id}")
def delete_org(org_id: int, _: User = Depends(get_admin_user)):
    from ..schemas import Org

    with get_session() as s:
        o = s.get(Org, org_id)
        if not o:
            raise HTTPException(404, "org not found")
        if s.query(User).filter_by(org_id=org_id).count()                 or s.query(Model).filter_by(org_id=org_id).count():
            raise HTTPException(409, "org still has users or models")
        ar_delete(s, o)
        return {"deleted": org_id}


@router.get("/clusters")
def list_clusters(_: User = Depends(get_current_user)):
    from ..schemas import Cluster

    with get_session() as s:
        items = []
        for c in s.query(Cluster).all():
            d = c.to_dict()
            d["workers"] = s.query(Worker).filter_by(cluster_id=c.id).count()
            items.append(d)
        return {"items": items}


@router.post("/clusters", status_code=201)
def create_cluster(body: ClusterCreate, _: User = Depends(get_admin_user)):
    """Creates the cluster AND a registration token scoped to it — workers
    registering with that token land in this cluster."""
    from ..schemas import Cluster
    from ..security import generate_registration_token

    with get_session() as s:
        if s.query(Cluster).filter_by(name=body.name).first():
            raise HTTPException(409, "cluster exists")
        c = Cluster(name=body.name, description=body.description)
        ar_create(s, c)
        tok = RegistrationToken(token=generate_registration_token(),
                                description=f"cluster {body.name}",
                                cluster_id=c.id)
        ar_create(s, tok)
        d = c.to_dict()
        d["registration_token"] = tok.token
        return d


@router.delete("/clusters/{cluster_id}")
def delete_cluster(cluster_id: int, _: User = Depends(get_admin_user)):
    from ..schemas import Cluster

    with get_session() as s:
        c = s.get(Cluster, cluster_id)
        if not c:
            raise HTTPException(404, "cluster not found")
        if c.is_default:
            raise HTTPException(400, "cannot delete the default cluster")
        if s.query(Worker).filter_by(cluster_id=cluster_id).count():
            raise HTTPException(409, "cluster still has workers")
        for t in s.query(RegistrationToken).filter_by(cluster_id=cluster_id).all():
            s.delete(t)
        s.commit()  # tokens must go before the cluster row (FK, no ORM rel)
        ar_delete(s, c)
        return {"deleted": cluster_id}


# -- GPU instances (operator-analog SSH GPU pods, server/gpu_instances.py) --

@router.get("/gpu_instances")
def list_gpu_instances(watch: bool = Query(False),
                       _: User = Depends(get_current_user)):
    from ..schemas import GPUInstance

    with get_session() as s:
        rows = [g.to_dict() for g in s.query(GPUInstance).all()]
    if watch:
        return _watch_stream("gpu_instances", rows, None)
    return {"items": rows}


@router.get("/gpu_instances/{gid}")
def get_gpu_instance(gid: int, _: User = Depends(get_current_user)):
    from ..schemas import GPUInstance

    with get_session() as s:
        g = s.get(GPUInstance, gid)
        if not g:
            raise HTTPException(404, "gpu instance not found")
        return g.to_dict()


@router.post("/gpu_instances", status_code=201)
def create_gpu_instance(body: GPUInstanceCreate,
                        _: User = Depends(get_admin_user)):
    from ..schemas import GPUInstance
    from .gpu_instances import FLAVORS, PROVIDERS

    from ..schemas import GPUInstanceTemplate, SSHPublicKey

    fields = {"flavor": body.flavor, "image": body.image,
              "volumes": body.volumes, "labels": body.labels,
              "provider": body.provider,
              "provider_config": body.provider_config}
    with get_session() as s:
        if body.template:
            t = (s.query(GPUInstanceTemplate)
                 .filter_by(name=body.template).first())
            if not t:
                raise HTTPException(400,
                                    f"unknown template {body.template!r}")
            # caller overrides beat template values; fields left at
            # their schema defaults inherit from the template
            set_fields = body.model_fields_set
            for f in fields:
                if f not in set_fields:
                    fields[f] = getattr(t, f)
        key = body.ssh_public_key
        if body.ssh_key_name:
            k = (s.query(SSHPublicKey)
                 .filter_by(name=body.ssh_key_name).first())
            if not k:
                raise HTTPException(400,
                                    f"unknown ssh key {body.ssh_key_name!r}")
            key = k.public_key
        if fields["provider"] not in PROVIDERS:
            raise HTTPException(400,
                                f"unknown provider {fields['provider']!r}")
        if fields["flavor"] not in FLAVORS:
            raise HTTPException(400, f"unknown flavor {fields['flavor']!r}; "
                                     f"one of {sorted(FLAVORS)}")
        if s.query(GPUInstance).filter_by(name=body.name).first():
            raise HTTPException(409, "gpu instance exists")
        g = GPUInstance(name=body.name, ssh_public_key=key, **fields)
        ar_create(s, g)
        return g.to_dict()


@router.delete("/gpu_instances/{gid}")
def delete_gpu_instance(gid: int, _: User = Depends(get_admin_user)):
    """Marks DELETING; the controller deprovisions the pod and removes
    the row (async, like instance teardown elsewhere)."""
    from ..schemas import GPUInstance, GPUInstanceState

    with get_session() as s:
        g = s.get(GPUInstance, gid)
        if not g:
            raise HTTPException(404, "gpu instance not found")
        g.state = GPUInstanceState.DELETING.value
        ar_update(s, g)
    return {"status": "deleting"}


# -- templates / ssh keys (reference: gpu_instance_templates,
# gpu_instance_ssh_public_keys) --

@router.get("/gpu_instance_templates")
def list_gpu_instance_templates(_: User = Depends(get_current_user)):
    from ..schemas import GPUInstanceTemplate

    with get_session() as s:
        return {"items": [t.to_dict()
                          for t in s.query(GPUInstanceTemplate).all()]}


@router.post("/gpu_instance_templates", status_code=201)
def create_gpu_instance_template(body: GPUInstanceTemplateCreate,
                                 _: User = Depends(get_admin_user)):
    from ..schemas import GPUInstanceTemplate
    from .gpu_instances import FLAVORS

    if body.flavor not in FLAVORS:
        raise HTTPException(400, f"unknown flavor {body.flavor!r}")
    with get_session() as s:
        if s.query(GPUInstanceTemplate).filter_by(name=body.name).first():
            raise HTTPException(409, "template exists")
        t = GPUInstanceTemplate(**body.model_dump())
        ar_create(s, t)
        return t.to_dict()


@router.delete("/gpu_instance_templates/{tid}")
def delete_gpu_instance_template(tid: int,
                                 _: User = Depends(get_admin_user)):
    from ..schemas import GPUInstanceTemplate

    with get_session() as s:
        t = s.get(GPUInstanceTemplate, tid)
        if not t:
            raise HTTPException(404, "template not found")
        ar_delete(s, t)
    return {"status": "deleted"}


@router.get("/ssh_public_keys")
def list_ssh_keys(user: User = Depends(get_current_user)):
    from ..schemas import SSHPublicKey

    with get_session() as s:
        q = s.query(SSHPublicKey)
        if not user.is_admin:
            q = q.filter_by(user_id=user.id)
        return {"items": [k.to_dict() for k in q.all()]}


@router.post("/ssh_public_keys", status_code=201)
def create_ssh_key(body: SSHPublicKeyCreate,
                   user: User = Depends(get_current_user)):
    from ..schemas import SSHPublicKey

    if not body.public_key.strip().startswith(("ssh-", "ecdsa-")):
        raise HTTPException(400, "not an SSH public key")
    with get_session() as s:
        if s.query(SSHPublicKey).filter_by(name=body.name).first():
            raise HTTPException(409, "key name exists")
        k = SSHPublicKey(name=body.name, user_id=user.id,
                         public_key=body.public_key.strip())
        ar_create(s, k)
        return k.to_dict()


@router.delete("/ssh_public_keys/{kid}")
def delete_ssh_key(kid: int, user: User = Depends(get_current_user)):
    from ..schemas import SSHPublicKey

    with get_session() as s:
        k = s.get(SSHPublicKey, kid)
        if not k or (not user.is_admin and k.user_id != user.id):
            raise HTTPException(404, "key not found")
        ar_delete(s, k)
    return {"status": "deleted"}


@router.get("/gpu_instance_flavors")
def list_gpu_instance_flavors(_: User = Depends(get_current_user)):
    from .gpu_instances import FLAVORS

    return {"items": [{"name": k, **v} for k, v in FLAVORS.items()]}


@router.get("/worker_pools")
def list_worker_pools(_: User = Depends(get_current_user)):
    from ..schemas import WorkerPool

    with get_session() as s:
        return {"items": [p.to_dict() for p in s.query(WorkerPool).all()]}


@router.post("/worker_pools", status_code=201)
def create_worker_pool(body: WorkerPoolCreate,
                       _: User = Depends(get_admin_user)):
    from ..schemas import WorkerPool
    from .providers import PROVIDERS

    if body.provider not in PROVIDERS:
        raise HTTPException(400, f"unknown provider {body.provider!r}")
    with get_session() as s:
        if s.query(WorkerPool).filter_by(name=body.name).first():
            raise HTTPException(409, "pool exists")
        p = WorkerPool(name=body.name, provider=body.provider,
                       instance_type=body.instance_type,
                       replicas=body.replicas,
                       provider_config=body.provider_config,
                       labels=body.labels, instances=[])
        ar_create(s, p)
        return p.to_dict()


@router.put("/worker_pools/{pool_id}")
def update_worker_pool(pool_id: int, body: WorkerPoolUpdate,
                       _: User = Depends(get_admin_user)):
    from ..schemas import WorkerPool

    with get_session() as s:
        p = s.get(WorkerPool, pool_id)
        if not p:
            raise HTTPException(404, "pool not found")
        for k, v in body.model_dump().items():
            if v is not None:
                setattr(p, k, v)
        ar_update(s, p)
        return p.to_dict()


@router.delete("/worker_pools/{pool_id}")
def delete_worker_pool(pool_id: int, _: User = Depends(get_admin_user)):
    from ..schemas import WorkerPool
    from .providers import get_provider

    with get_session() as s:
        p = s.get(WorkerPool, pool_id)
        if not p:
            raise HTTPException(404, "pool not found")
        try:
            provider = get_provider(p.provider, p.provider_config)
            for r in p.instances or []:
                provider.delete(r["instance_id"])
        except Exception:  # noqa: BLE001
            pass  # deprovision is best-effort on delete
        ar_delete(s, p)
        return {"deleted": pool_id}


@router.get("/model_routes")
def list_routes(_: User = Depends(get_current_user)):
    with get_session() as s:
        return {"items": [r.to_dict() for r in s.query(ModelRoute).all()]}


@router.post("/model_routes", status_code=201)
def create_route(body: ModelRouteCreate, _: User = Depends(get_current_user)):
    with get_session() as s:
        if s.query(ModelRoute).filter_by(name=body.name).first():
            raise HTTPException(409, "route exists")
        r = ModelRoute(name=body.name, targets=body.targets)
        ar_create(s, r)
        return r.to_dict()


@router.delete("/model_routes/{route_id}")
def delete_route(route_id: int, _: User = Depends(get_current_user)):
    with get_session() as s:
        r = s.get(ModelRoute, route_id)
        if not r:
            raise HTTPException(404)
        ar_delete(s, r)
        return {"ok": True}


# ---- model evaluations (compatibility pre-check) ---------------------------

# evaluation results are deterministic for a given (model spec, worker set);
# cache them briefly (reference: scheduler/evaluator.py TTL result cache) and
# invalidate cluster-wide when workers change (see worker routes).
_eval_cache = None


def get_eval_cache():
    global _eval_cache
    if _eval_cache is None:
        from .cache import TTLCache

        _eval_cache = TTLCache(ttl=30.0, maxsize=256,
                               distributed_name="model_evaluations")
    return _eval_cache


@router.post("/model-evaluations")
def evaluate_model(body: ModelCreate, _: User = Depends(get_current_user)):
    """Dry-run of the placement pipeline (reference: scheduler/evaluator.py):
    reports whether the model would fit, on which worker, and why not."""
    from ..scheduler.policies import (
        estimate_vram_claim, model_spec_for, pick_candidate,
    )

    cache = get_eval_cache()
    ck = (body.model_ref, body.source, body.gpus_per_replica,
          body.gpu_memory_utilization)
    hit = cache.get(ck)
    if hit is not None:
        return hit

    model_d = body.model_dump() | {"id": -1}
    spec = model_spec_for(model_d)
    if spec is None:
        return {"compatible": False,
                "messages": [f"cannot resolve model spec for {body.model_ref!r}"]}
    tp = max(1, body.gpus_per_replica)
    claim = estimate_vram_claim(model_d, spec, tp)
    with get_session() as s:
        workers = [w.to_dict() for w in s.query(Worker).all()]
        insts = [i.to_dict() for i in s.query(ModelInstance).all()]
    cand = pick_candidate(model_d, workers, insts)
    if cand is None:
        result = {
            "compatible": False,
            "estimated_vram_per_gpu": claim,
            "messages": ["no worker currently fits the resource claim"],
        }
        cache.set(ck, result)
        return result
    result = {
        "compatible": True,
        "estimated_vram_per_gpu": claim,
        "candidate": {"worker": cand.worker.get("name"),
                      "gpu_indexes": cand.gpu_indexes},
        "messages": [],
    }
    cache.set(ck, result)
    return result


# ---- benchmarks ------------------------------------------------------------

@router.get("/benchmarks")
def list_benchmarks(watch: bool = Query(False), user: User = Depends(get_current_user)):
    with get_session() as s:
        rows = [b.to_dict() for b in s.query(Benchmark).all()]
    if watch:
        return _watch_stream("benchmarks", rows, None)
    return {"items": rows}


@router.post("/benchmarks", status_code=201)
def create_benchmark(body: BenchmarkCreate, _: User = Depends(get_current_user)):
    with get_session() as s:
        model = s.query(Model).filter_by(name=body.model_name).first()
        if not model:
            raise HTTPException(404, f"model {body.model_name!r} not found")
        inst = (
            s.query(ModelInstance)
            .filter_by(model_id=model.id, state=ModelInstanceState.RUNNING.value)
            .first()
        )
        if not inst:
            raise HTTPException(409, "model has no running instance to benchmark")
        b = Benchmark(
            name=body.name, model_name=body.model_name, worker_id=inst.worker_id,
            config={"mode": body.mode, "value": body.value,
                    "sweep": body.sweep,
                    "duration_s": body.duration_s, "isl": body.isl, "osl": body.osl},
        )
        ar_create(s, b)
        return b.to_dict()


@router.patch("/benchmarks/{bench_id}")
def update_benchmark(bench_id: int, body: dict, _=Depends(verify_worker_token)):
    with get_session() as s:
        b = s.get(Benchmark, bench_id)
        if not b:
            raise HTTPException(404)
        for k in ("state", "state_message", "results"):
            if k in body:
                setattr(b, k, body[k])
        ar_update(s, b)
        return b.to_dict()


@router.delete("/benchmarks/{bench_id}")
def delete_benchmark(bench_id: int, _: User = Depends(get_current_user)):
    with get_session() as s:
        b = s.get(Benchmark, bench_id)
        if not b:
            raise HTTPException(404)
        ar_delete(s, b)
        return {"ok": True}


# ---- tunnel proxy (NAT workers; reference websocket_proxy/) ----------------

@router.get("/tunnel/jobs")
async def tunnel_jobs(worker_id: int, batch: int = 0,
                      _=Depends(verify_worker_token)):
    """Long-poll job pickup. `batch=1` returns {"jobs": [...]} with every
    queued job (first blocks, rest drain) so a burst of concurrent
    requests does not pay one poll round-trip each."""
    from .tunnel import hub

    if batch:
        return {"jobs": await hub.next_jobs(worker_id)}
    job = await hub.next_job(worker_id)
    if job is None:
        from fastapi.responses import Response as _Resp

        return _Resp(status_code=204)
    return job


@router.post("/tunnel/reply/{req_id}")
async def tunnel_reply(req_id: str, request: Request, _=Depends(verify_worker_token)):
    from .tunnel import hub

    status = int(request.headers.get("x-tunnel-status", "502"))
    ctype = request.headers.get("x-tunnel-content-type", "application/json")
    q = await hub.begin_reply(req_id, status, ctype)
    if q is None:
        raise HTTPException(410, "request no longer waiting")
    async for chunk in request.stream():
        if chunk:
            await hub.push_chunk(req_id, chunk)
    await hub.end_reply(req_id)
    return {"ok": True}


# ---- model providers (external OpenAI-compatible backends) -----------------

@router.get("/model_providers")
def list_providers(_: User = Depends(get_current_user)):
    with get_session() as s:
        return {"items": [p.to_dict() | {"api_key": "***" if p.api_key else ""}
                          for p in s.query(ModelProvider).all()]}


@router.post("/model_providers", status_code=201)
def create_provider(body: ModelProviderCreate, _: User = Depends(get_admin_user)):
    with get_session() as s:
        if s.query(ModelProvider).filter_by(name=body.name).first():
            raise HTTPException(409, "provider exists")
        p = ModelProvider(**body.model_dump())
        ar_create(s, p)
        return p.to_dict() | {"api_key": "***" if p.api_key else ""}


@router.delete("/model_providers/{provider_id}")
def delete_provider(provider_id: int, _: User = Depends(get_admin_user)):
    with get_session() as s:
        p = s.get(ModelProvider, provider_id)
        if not p:
            raise HTTPException(404)
        ar_delete(s, p)
        return {"ok": True}


# ---- catalog / version -----------------------------------------------------

@router.get("/catalog")
def catalog(_: User = Depends(get_current_user)):
    """Built-in model catalog (reference: server/catalog.py)."""
    import json as _json
    from pathlib import Path as _P

    return _json.loads((_P(__file__).parent / "catalog.json").read_text())


@router.get("/version")
def version():
    from .. import __version__

    return {"version": __version__}


@router.get("/model_instances/{instance_id}/logs")
async def instance_logs(instance_id: int, tail: int = Query(200),
                        _: User = Depends(get_current_user)):
    """Instance log tail proxied through the server (reference:
    routes/model_instances `logs` → worker log API), honoring the
    worker's proxy mode (direct HTTP or the NAT tunnel)."""
    from fastapi.responses import PlainTextResponse

    with get_session() as s:
        inst = s.get(ModelInstance, instance_id)
        if not inst:
            raise HTTPException(404, "instance not found")
        w = s.get(Worker, inst.worker_id) if inst.worker_id else None
    if w is None:
        raise HTTPException(409, "instance has no worker yet")
    path = f"/logs/{inst.name}?tail={int(tail)}"
    if w.proxy_mode == "tunnel":
        from .tunnel import hub

        try:
            status, _ctype, chunks = await hub.request(
                w.id, w.port, "GET", path, b"")
        except TimeoutError:
            raise HTTPException(502, "tunnel reply timeout")
        buf = b""
        async for c in chunks:
            buf += c
        return PlainTextResponse(buf.decode(errors="replace"),
                                 status_code=status)
    import httpx as _httpx

    url = f"http://{w.ip}:{w.port}{path}"
    try:
        async with _httpx.AsyncClient(timeout=15) as c:
            r = await c.get(url)
    except _httpx.HTTPError:
        raise HTTPException(502, "worker unreachable")
    return PlainTextResponse(r.text, status_code=r.status_code)


@router.get("/gpu_devices")
def list_gpu_devices(page: int | None = Query(None),
                     perPage: int | None = Query(None),
                     search: str | None = Query(None),
                     _: User = Depends(get_current_user)):
    """Flattened GPU inventory across workers (reference:
    routes/gpu_devices.py): device info + per-GPU allocatable VRAM from
    the same claim accounting the scheduler places against."""
    from ..scheduler.policies import worker_allocatable

    with get_session() as s:
        workers = [w.to_dict() for w in s.query(Worker).all()]
        insts = [i.to_dict() for i in s.query(ModelInstance).all()]
    rows = []
    for w in workers:
        alloc = worker_allocatable(w, insts)
        gpus = ((w.get("status") or {}).get("gpu_devices")
                or w.get("gpu_devices") or [])
        for g in gpus:
            idx = g.get("index", 0)
            rows.append({
                "id": f"{w['name']}:{idx}",
                "worker_id": w["id"], "worker_name": w["name"],
                "index": idx, "name": g.get("name", ""),
                "type": g.get("type", "rocm"),
                "memory": g.get("memory") or {},
                "allocatable_vram": alloc.get(idx, 0),
                "compute_partition": g.get("compute_partition"),
                "memory_partition": g.get("memory_partition"),
            })
    return _paginate(rows, page, perPage, search)


@router.get("/config")
def server_config(request: Request, _: User = Depends(get_admin_user)):
    """Sanitized effective server config (reference routes/config.py):
    secrets and tokens are withheld."""
    cfg = request.app.state.config
    hide = {"token", "jwt_secret", "bootstrap_password", "database_url"}
    out = {}
    for k, v in vars(cfg).items():
        if k.startswith("_"):
            continue
        out[k] = "***" if (k in hide and v) else v
    return out


# ---- dashboard / usage / system load ---------------------------------------

@router.get("/dashboard")
def dashboard(_: User = Depends(get_current_user)):
    """Aggregate operator view (reference: routes/dashboard.py):
    resource counts, current + recent system load, and a 7-day model
    usage summary with per-model totals."""
    import datetime as _dt

    from ..schemas import GPUInstance, ModelInstanceState

    with get_session() as s:
        workers = s.query(Worker).all()
        gpu_count = sum(len(w.gpu_devices or []) for w in workers)
        vram_total = sum((g.get("memory") or {}).get("total", 0)
                         for w in workers for g in (w.gpu_devices or []))
        models = s.query(Model).count()
        insts = s.query(ModelInstance).all()
        counts = {
            "workers": len(workers),
            "gpus": gpu_count,
            "vram_total_bytes": vram_total,
            "models": models,
            "model_instances": len(insts),
            "running_instances": sum(
                1 for i in insts
                if i.state == ModelInstanceState.RUNNING.value),
            "gpu_instances": s.query(GPUInstance).count(),
        }
        loads = (s.query(SystemLoad)
                 .order_by(SystemLoad.timestamp.desc()).limit(60).all())
        current = loads[0].to_dict() if loads else None
        history = [r.to_dict() for r in reversed(loads)]
        since = (_dt.date.today()
                 - _dt.timedelta(days=6)).strftime("%Y-%m-%d")
        rows = s.query(ModelUsage).filter(ModelUsage.date >= since).all()
        per_model: dict[str, dict] = {}
        totals = {"prompt_tokens": 0, "completion_tokens": 0,
                  "request_count": 0}
        for u in rows:
            m = per_model.setdefault(u.model_name, {
                "model_name": u.model_name, "prompt_tokens": 0,
                "completion_tokens": 0, "request_count": 0})
            for k in totals:
                v = getattr(u, k) or 0
                m[k] += v
                totals[k] += v
        top = sorted(per_model.values(),
                     key=lambda m: -(m["prompt_tokens"]
                                     + m["completion_tokens"]))[:10]
    return {
        "resource_counts": counts,
        "system_load": {"current": current, "history": history},
        "model_usage": {"since": since, "totals": totals,
                        "top_models": top},
    }



@router.get("/usage")
def usage(user: User = Depends(get_current_user)):
    with get_session() as s:
        q = s.query(ModelUsage)
        if not user.is_admin:
            q = q.filter_by(user_id=user.id)
        return {"items": [u.to_dict() for u in q.all()]}


@router.get("/system_load")
def system_load(_: User = Depends(get_current_user)):
    with get_session() as s:
        rows = s.query(SystemLoad).order_by(SystemLoad.timestamp.desc()).limit(120).all()
        return {"items": [r.to_dict() for r in rows]}
