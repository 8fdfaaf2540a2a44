"""Serving controller: model registry + HTTP frontend.

Capability analog of the reference's ``alpa/serve/controller.py``
(Controller:96 + DeviceMeshGroupManager replicas + uvicorn/starlette HTTP
proxy).  Without Ray, the controller runs in the rank-0 process of the
serving job: it owns a registry of generate callables (each backed by a
TP-sharded model on the local mesh) and a starlette ASGI app; non-zero
ranks sit in a broadcast-driven worker loop executing the same generate
calls (SPMD serving).
"""
from __future__ import annotations

import json
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

import torch

from ..mesh import is_distributed, rank


@dataclass
class CreateInfo:
    """Model registration record (reference CreateInfo, controller.py:35)."""
    name: str
    generate_fn: Callable  # (prompt_ids [B,S] LongTensor, max_tokens) -> ids


class Controller:

    def __init__(self):
        self.models: Dict[str, CreateInfo] = {}

    def register_model(self, name: str, generate_fn: Callable):
        self.models[name] = CreateInfo(name, generate_fn)

    def list_models(self) -> List[str]:
        return sorted(self.models)

    def completions(self, model: str, prompt_ids: List[List[int]],
                    max_tokens: int = 16, num_beams: int = 1
                    ) -> List[List[int]]:
        info = self.models[model]
        ids = torch.tensor(prompt_ids, dtype=torch.long)
        if num_beams > 1:
            out = info.generate_fn(ids, max_tokens, num_beams=num_beams)
        else:
            out = info.generate_fn(ids, max_tokens)
        return out.tolist()

    # ------------------------- HTTP frontend -------------------------
    def asgi_app(self):
        from starlette.applications import Starlette
        from starlette.responses import JSONResponse
        from starlette.routing import Route

        async def models(request):
            return JSONResponse({"models": self.list_models()})

        async def completions(request):
            body = await request.json()
            try:
                out = self.completions(
                    body["model"], body["prompt_ids"],
                    int(body.get("max_tokens", 16)),
                    int(body.get("num_beams", 1)))
            except KeyError as e:
                return JSONResponse({"error": f"missing field {e}"},
                                    status_code=400)
            return JSONResponse({"output_ids": out})

        return Starlette(routes=[
            Route("/models", models, methods=["GET"]),
            Route("/completions", completions, methods=["POST"]),
        ])


def run_controller(controller: Controller, host: str = "127.0.0.1",
                   port: int = 8265):
    """Blocking uvicorn server (reference run_controller,
    controller.py:280)."""
    import uvicorn
    uvicorn.run(controller.asgi_app(), host=host, port=port,
                log_level="warning")


# ---------------------------------------------------------------------------
# SPMD serving: rank 0 owns HTTP, other ranks execute the same generate
# calls (reference: DeviceMeshGroupManager replica actors execute driver
# RPCs, controller.py:59)
# ---------------------------------------------------------------------------


def _bcast(obj):
    import torch.distributed as dist
    lst = [obj]
    dist.broadcast_object_list(lst, src=0)
    return lst[0]


class SpmdGenerateService:
    """Wraps a generate fn so every rank of a TP group runs it in
    lockstep: rank 0 (driving HTTP) broadcasts each request's inputs;
    other ranks sit in `serve_worker_loop` executing them.  A `None`
    broadcast shuts the workers down."""

    def __init__(self, generate_fn):
        from ..mesh import is_distributed, rank
        self.generate_fn = generate_fn
        self._dist = is_distributed()
        self._is_driver = rank() == 0

    def __call__(self, ids, max_tokens, num_beams: int = 1):
        assert self._is_driver
        if self._dist:
            _bcast(("gen", ids.cpu(), int(max_tokens), int(num_beams)))
        return self._run(ids, max_tokens, num_beams)

    def _run(self, ids, max_tokens, num_beams):
        if num_beams > 1:
            return self.generate_fn(ids, max_tokens, num_beams=num_beams)
        return self.generate_fn(ids, max_tokens)

    def shutdown_workers(self):
        if self._dist and self._is_driver:
            _bcast(None)

    def serve_worker_loop(self):
        """Non-zero ranks: execute broadcast requests until shutdown."""
        assert not self._is_driver
        while True:
            msg = _bcast(None)
            if msg is None:
                return
            _, ids, max_tokens, num_beams = msg
            self._run(ids, max_tokens, num_beams)
