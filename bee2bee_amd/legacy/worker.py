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
        if kind == protocol.TASK_ONNX_LOAD:
            # gated on onnxruntime like the reference (bee2bee/node.py:210-220)
            try:
                import onnxruntime as ort  # noqa: F401
            except Exception:
                raise RuntimeError("onnx_support_missing")
            model_id = payload.get("model_id") or new_id("onnx")
            sess = ort.InferenceSession(payload["path"])
            self._engines[model_id] = {"type": "onnx", "session": sess}
            return {"model_id": model_id}
        if kind == protocol.TASK_ONNX_INFER:
            m = self._engines.get(payload.get("model_id"))
            if not isinstance(m, dict) or m.get("type") != "onnx":
                raise RuntimeError("onnx_model_not_loaded")
            inputs = payload.get("inputs") or {}
            out = m["session"].run(
                None, {k: np.array(v) for k, v in inputs.items()}
            )
            return {"outputs": [o.tolist() if hasattr(o, "tolist") else o
                                for o in out]}
        if kind == protocol.TASK_ONNX_UNLOAD:
            self._engines.pop(payload.get("model_id"), None)
            return {"ok": True}
        if kind == protocol.TASK_HF_PART_LOAD:
            # layer-range partial of OUR transformer (the reference's
            # DistilBERT partial, bee2bee/node.py:236-249, rebuilt on the
            # native stack: same math as a pipeline stage)
            import torch

            from ..engine.kv import PagedKV
            from ..engine.runner import Runner
            from ..models.spec import resolve_spec
            from ..models.tokenizer import load_tokenizer
            from ..models.weights import ModelWeights

            name = payload.get("model_name", "tiny")
            start = int(payload.get("start", 0))
            end = int(payload.get("end", 2))
            model_id = payload.get("model_id") or new_id("hfpart")
            spec = resolve_spec(name, payload.get("model_path"))
            end = min(end, spec.n_layers)
            dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
            dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
            w = ModelWeights(spec, dev, dtype)
            if payload.get("model_path"):
                w.load_hf(payload["model_path"], layer_range=(start, end))
            else:
                w.random_init(seed=int(payload.get("seed", 0)),
                              layer_range=(start, end))
            kv = PagedKV(spec, dev, dtype, n_blocks=32,
                         layer_range=(start, end))
            runner = Runner(spec, w, kv, dev, dtype,
                            layer_range=(start, end))
            tok = load_tokenizer(payload.get("model_path"), spec.vocab_size,
                                 spec.bos_token_id, spec.eos_token_id)
            self._engines[model_id] = {
                "type": "hf_part", "runner": runner, "kv": kv, "tok": tok,
                "start": start, "end": end, "seq": 0,
            }
            return {"model_id": model_id, "start": start, "end": end}
        if kind == protocol.TASK_HF_PART_FORWARD:
            import torch

            m = self._engines.get(payload.get("model_id"))
            if not isinstance(m, dict) or m.get("type") != "hf_part":
                raise RuntimeError("model_not_loaded")
            runner, kv = m["runner"], m["kv"]
            dev = runner.device
            if payload.get("text") is not None:
                ids = m["tok"].encode(payload["text"])
                inp = torch.tensor(ids, dtype=torch.int64, device=dev)
            else:
                hs = np.array(payload["hidden"], dtype=np.float32)
                inp = torch.from_numpy(hs).to(dev, runner.dtype)
            T = inp.shape[0]
            sid = m["seq"]
            m["seq"] += 1
            kv.new_seq(sid)
            kv.extend_seq(sid, T)
            slots = torch.tensor(kv.slot_mapping(sid, range(T)),
                                 dtype=torch.int32, device=dev)
            pos = torch.arange(T, dtype=torch.int32, device=dev)
            cu = torch.tensor([0, T], dtype=torch.int32, device=dev)
            with torch.no_grad():
                hid = runner.forward_prefill(inp, pos, slots, cu, T)
            kv.free_seq(sid)
            return {"hidden": hid.float().cpu().numpy().tolist()}
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
