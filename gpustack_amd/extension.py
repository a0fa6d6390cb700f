"""Extension plugins (reference: gpustack/extension.py — entry-point group
`gpustack.plugins` can mount routers, add CLI flags and supply the HA
Coordinator).

First-party equivalent on the entry-point group `gpustack_amd.plugins`.
A plugin is any object (usually a module) exposing some of:

    def routers() -> list[fastapi.APIRouter]        # mounted on the server
    def coordinator(cfg) -> object | None           # replaces the HA coordinator
    def on_server_start(app, cfg) -> None           # startup hook
"""
from __future__ import annotations

import logging

logger = logging.getLogger(__name__)


def load_plugins() -> list:
    try:
        from importlib.metadata import entry_points
    except ImportError:  # pragma: no cover
        return []
    plugins = []
    try:
        eps = entry_points(group="gpustack_amd.plugins")
    except TypeError:  # older importlib.metadata API
        eps = entry_points().get("gpustack_amd.plugins", [])
    for ep in eps:
        try:
            plugins.append(ep.load())
            logger.info("loaded plugin %s", ep.name)
        except Exception:  # noqa: BLE001
            logger.exception("plugin %s failed to load", ep.name)
    return plugins


def apply_routers(app, plugins) -> int:
    n = 0
    for p in plugins:
        for router in (getattr(p, "routers", lambda: [])() or []):
            app.include_router(router)
            n += 1
    return n


def pick_coordinator(cfg, plugins):
    for p in plugins:
        fn = getattr(p, "coordinator", None)
        if fn:
            c = fn(cfg)
            if c is not None:
                return c
    return None


def run_start_hooks(app, cfg, plugins) -> None:
    for p in plugins:
        fn = getattr(p, "on_server_start", None)
        if fn:
            try:
                fn(app, cfg)
            except Exception:  # noqa: BLE001
                logger.exception("plugin start hook failed")
