"""Coordinator-protocol worker: connects to a coordinator WS URL, registers
its resources, and executes layer/HF tasks (reference bee2bee/node.py:48-293
behavior; auto-reconnect loop included).

The numpy layer tasks run the legacy MLP math; the HF tasks are served by
the native MI355X engine (hf_load builds an InferenceEngine, hf_infer
generates). Tensor payloads stay JSON float lists for wire compatibility.
"""
from __future__ import annotations

import asyncio
import json
import logging
import platform
from typing import Any, Dict, Optional

import aiohttp
import numpy as np

from ..utils import get_system_metrics, new_id
from . import mlp, protocol

logger = logging.getLogger("bee2bee_amd.legacy")


def gather_resources() -> Dict[str, Any]:
    import psutil

    try:
        import torch

        gpu = torch.cuda.is_available()
    except Exception:
        gpu = False
    return {
        "cpu_count": psutil.cpu_count(),
        "ram_gb": round(psutil.virtual_memory().total / 2**30, 1),
        "gpu": gpu,
        "platform": platform.system(),
    }


class LegacyWorker:
    def __init__(self, name: str = "worker", price: float = 0.0) -> None:
        self.node_id = new_id("node")
        self.name = name
        self.price = price
        self._fwd_cache: Dict[str, Any] = {}
        self._engines: Dict[str, Any] = {}

    # ------------------------------------------------------------ task exec

    def execute_task(self, kind: str, payload: Dict[str, Any]) -> Dict[str, Any]:
        if kind == protocol.TASK_LAYER_FORWARD:
            layer = mlp.Layer.from_json(payload["layer"])
            x = np.asarray(payload["x"], dtype=np.float64)
            return {"y": mlp.layer_forward(layer, x).tolist()}
        if kind == protocol.TASK_LAYER_FORWARD_TRAIN:
            layer = mlp.Layer.from_json(payload["layer"])
            x = np.asarray(payload["x"], dtype=np.float64)
            z = x @ layer.w + layer.b
            cache_id = payload.get("cache_id") or new_id("cache")
            self._fwd_cache[cache_id] = (layer, x, z)
            y = mlp.layer_forward(layer, x)
            return {"y": y.tolist(), "cache_id": cache_id}
        if kind == protocol.TASK_LAYER_BACKWARD:
            cache_id = payload["cache_id"]
            layer, x, z = self._fwd_cache.pop(cache_id)
            grad = np.asarray(payload["grad"], dtype=np.float64)
            dx, gw, gb = mlp.layer_backward(layer, x, z, grad)
            return {"dX": dx.tolist(), "gW": gw.tolist(), "gb": gb.tolist()}
        if kind == protocol.TASK_HF_LOAD:
            from ..engine.engine import InferenceEngine

            name = payload["model"]
            if name not in self._engines:
                self._engines[name] = InferenceEngine(
                    name, model_path=payload.get("model_path"), max_batch=4
                )
                self._engines[name].start()
            return {"loaded": name}
        if kind == protocol.TASK_HF_INFER:
            eng = self._engines[payload["model"]]
            res = eng.generate_text(
                payload.get("prompt", ""),
                max_new_tokens=int(payload.get("max_new_tokens", 32)),
                temperature=float(payload.get("temperature", 0.7)),
            )
            return {"text": res["text"], "tokens": res["tokens"]}
        if kind == protocol.TASK_HF_UNLOAD:
            eng = self._engines.pop(payload["model"], None)
            if eng is not None:
                eng.shutdown()
            return {"unloaded": payload["model"]}
        raise ValueError(f"unsupported task kind: {kind}")

    # ------------------------------------------------------------ transport

    async def run(self, coordinator_url: str, reconnect_s: float = 2.0,
                  once: bool = False) -> None:
        while True:
            try:
                await self._session_loop(coordinator_url)
            except Exception as e:
                logger.warning("coordinator link lost: %s", e)
            if once:
                return
            await asyncio.sleep(reconnect_s)

    async def _session_loop(self, url: str) -> None:
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(url) as ws:
                await ws.send_str(json.dumps(protocol.msg(
                    protocol.MSG_REGISTER,
                    node_id=self.node_id,
                    name=self.name,
                    resources=gather_resources(),
                    price=self.price,
                    metrics=get_system_metrics(),
                )))
                async for m in ws:
                    if m.type != aiohttp.WSMsgType.TEXT:
                        break
                    data = json.loads(m.data)
                    if data.get("type") != protocol.MSG_TASK:
                        continue
                    task_id = data.get("task_id")
                    try:
                        result = self.execute_task(
                            data.get("kind"), data.get("payload", {})
                        )
                        await ws.send_str(json.dumps(protocol.msg(
                            protocol.MSG_RESULT, task_id=task_id, payload=result
                        )))
                    except Exception as e:  # noqa: BLE001
                        await ws.send_str(json.dumps(protocol.msg(
                            protocol.MSG_ERROR, task_id=task_id, message=str(e)
                        )))


def run_worker(coordinator_url: str, name: str = "worker") -> None:
    asyncio.run(LegacyWorker(name=name).run(coordinator_url))
