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
