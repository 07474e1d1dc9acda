"""Server bootstrap: wires DB, engine, agent loops, memory, task runner,
runtime loops, and the HTTP/WS app (reference boot path: cli/index.ts →
server/index.ts startServer → runtime.ts startServerRuntime).

Agents do NOT auto-resume on boot (reference runtime.ts:335-337); rooms start
via POST /api/rooms/{id}/start.
"""
from __future__ import annotations

import os

from ..core.agent_loop import AgentLoopManager
from ..core.events import EventBus
from ..core.tasks import TaskRunner
from ..db import LockedDb, connect
from ..memory.vector_store import GpuVectorStore, MemoryService
from .app import create_app
from .auth import AuthManager, data_dir
from .runtime import ServerRuntime


def default_db_path() -> str:
    return os.environ.get("ROOMAMD_DB_PATH", str(data_dir() / "data.db"))


def build_server(db_path: str | None = None, use_gpu: bool | None = None,
                 skip_token_file: bool = False):
    """Returns (app, components dict). Call `await components['runtime'].start()`
    inside the running loop (uvicorn startup hook does this)."""
    import torch

    if use_gpu is None:
        use_gpu = torch.cuda.is_available()
    ldb = LockedDb(connect(db_path or default_db_path()))
    bus = EventBus()
    auth = AuthManager(skip_token_file=skip_token_file)

    memory = MemoryService(ldb, store=GpuVectorStore(
        capacity=int(os.environ.get("ROOMAMD_VECTOR_CAPACITY", "1000000")),
        device="cuda" if use_gpu else "cpu"))
    with ldb as db:
        memory.store.rebuild_from_db(db)

    default_model = "qwen3-coder-30b" if use_gpu else "stub"
    if use_gpu:
        # engine is created lazily on first chat; warm it at boot so the first
        # cycle doesn't pay weight-init latency
        from ..engine.llm import get_local_engine
        from ..engine.providers import register_engine
        eng = get_local_engine(default_model)
        register_engine(default_model, eng)

    mgr = AgentLoopManager(ldb, bus=bus, memory=memory)
    runner = TaskRunner(ldb, bus=bus, memory=memory, default_model=default_model)
    runtime = ServerRuntime(ldb, runner, loop_mgr=mgr, memory=memory, bus=bus)
    app = create_app(ldb, loop_mgr=mgr, runner=runner, memory=memory,
                     auth=auth, bus=bus)

    # lifespan wiring (on_event is deprecated): runtime loops start with
    # the app and stop before the process exits
    import contextlib

    @contextlib.asynccontextmanager
    async def _lifespan(_app):
        await runtime.start()
        try:
            yield
        finally:
            await runtime.stop()
            mgr.stop_all()

    app.router.lifespan_context = _lifespan

    components = {"ldb": ldb, "bus": bus, "auth": auth, "memory": memory,
                  "loop_mgr": mgr, "runner": runner, "runtime": runtime}
    return app, components


def serve(port: int = 3700, host: str = "127.0.0.1",
          db_path: str | None = None) -> None:
    import uvicorn

    app, components = build_server(db_path=db_path)
    # register the MCP server into AI-client configs that exist on this
    # machine (reference index.ts:729-864; ROOMAMD_SKIP_MCP_REGISTER=1 skips)
    from .mcp_register import register_mcp_globally
    patched = register_mcp_globally(db_path or default_db_path())
    hit = [k for k, v in patched.items() if v]
    if hit:
        print(f"MCP registered into: {', '.join(hit)}")
    # persist the port for MCP cross-process nudges (reference: api.port file)
    try:
        (data_dir() / "api.port").write_text(str(port))
    except OSError:
        pass
    print(f"room_amd serving on http://{host}:{port} "
          f"(agent token at {data_dir() / 'api.token'})")
    uvicorn.run(app, host=host, port=port, log_level="warning")
