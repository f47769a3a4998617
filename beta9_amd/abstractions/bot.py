"""Bot: petri-net agent orchestration (experimental).

Reference: pkg/abstractions/experimental/bot/ — a bot is a network of
LOCATIONS (typed marker stores) and TRANSITIONS (container workloads that
fire when their input locations hold enough markers; their results are
pushed as markers to output locations). The reference layers an LLM chat
loop on top (types.go:155-230 BotConfig/BotTransitionConfig,
bot.proto PushBotMarkers/PushBotEvent/PopBotTask).

MI355X-native build: each transition is registered as a real function stub
at session-create time, so a firing transition runs through the normal
scheduler -> worker -> container path (reference parity: "each transition a
container"); the net engine lives in the gateway and reacts to marker
pushes event-driven (no polling). The optional conversational layer can be
served by beta9_amd.integrations.llm.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, Field

from ..rpc.channel import not_found, rpc
from ..types import StubConfig, TaskPolicy, new_id

log = logging.getLogger(__name__)


class BotLocationConfig(BaseModel):
    name: str = ""
    marker: Dict[str, str] = Field(default_factory=dict)  # field -> type name
    expose: bool = True


class BotTransitionConfig(BaseModel):
    name: str = ""
    handler: str = ""  # "module:function"
    inputs: Dict[str, int] = Field(default_factory=dict)  # location -> count
    outputs: List[str] = Field(default_factory=list)
    cpu: int = 1000
    memory: int = 1024
    gpu: str = ""
    timeout: float = 600.0
    description: str = ""
    expose: bool = True
    confirm: bool = False  # require an explicit event before firing


class BotConfig(BaseModel):
    model: str = ""  # optional LLM model for the chat layer
    locations: Dict[str, BotLocationConfig] = Field(default_factory=dict)
    transitions: Dict[str, BotTransitionConfig] = Field(default_factory=dict)
    welcome_message: str = ""


class BotSession:
    """One live petri net: markers per location + event ring + fired tasks."""

    def __init__(self, session_id: str, cfg: BotConfig, stub_ids: Dict[str, str]):
        self.session_id = session_id
        self.cfg = cfg
        self.transition_stubs = stub_ids  # transition name -> function stub id
        self.markers: Dict[str, List[Dict[str, Any]]] = {n: [] for n in cfg.locations}
        self.events: List[Dict[str, Any]] = []
        self.pending_confirm: Dict[str, bool] = {}
        self._event_waiter = asyncio.Event()

    def emit(self, typ: str, **fields) -> None:
        self.events.append({"type": typ, "ts": time.time(), **fields})
        if len(self.events) > 1000:
            self.events = self.events[-1000:]
        self._event_waiter.set()

    def satisfied(self, t: BotTransitionConfig) -> bool:
        return all(len(self.markers.get(loc, [])) >= n for loc, n in t.inputs.items())


class BotService:
    service_name = "BotService"

    def __init__(self, gateway):
        self.gw = gateway
        self.sessions: Dict[str, BotSession] = {}

    # -- session lifecycle ---------------------------------------------------
    @rpc()
    async def CreateBotSession(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        cfg = BotConfig.model_validate(req.get("config") or {})
        ws = auth["workspace_id"]
        app = self.gw.db.get_or_create_app(ws, req.get("name", "bot"))
        stub_ids: Dict[str, str] = {}
        for name, t in cfg.transitions.items():
            scfg = StubConfig(
                handler=t.handler,
                task_policy=TaskPolicy(max_retries=0, timeout=t.timeout),
                keep_warm_seconds=float(req.get("keep_warm_seconds", 10.0)),
                timeout=t.timeout,
            )
            scfg.runtime.cpu = t.cpu
            scfg.runtime.memory_mb = t.memory
            scfg.runtime.gpu = t.gpu
            if t.gpu:
                scfg.runtime.gpu_count = 1
            stub = self.gw.db.get_or_create_stub(
                ws, f"bot-{req.get('name', 'bot')}-{name}", "function", scfg, app_id=app["id"]
            )
            stub_ids[name] = stub["id"]
        session = BotSession(new_id("bot"), cfg, stub_ids)
        session.workspace_id = ws
        self.sessions[session.session_id] = session
        if cfg.welcome_message:
            session.emit("welcome", message=cfg.welcome_message)
        return {"session_id": session.session_id, "transition_stubs": stub_ids}

    def _session(self, sid: str) -> BotSession:
        s = self.sessions.get(sid)
        if s is None:
            raise not_found(f"bot session {sid}")
        return s

    # -- markers / events (reference: bot.proto PushBotMarkers/PushBotEvent)
    @rpc()
    async def PushBotMarkers(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        s = self._session(req["session_id"])
        for loc, markers in (req.get("markers") or {}).items():
            if loc not in s.markers:
                return {"ok": False, "error": f"unknown location {loc}"}
            for m in markers:
                s.markers[loc].append(dict(m))
                s.emit("marker_pushed", location=loc, marker=m)
        await self._fire_ready(s)
        return {"ok": True}

    @rpc()
    async def PushBotEvent(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """User events; `confirm_transition` releases a Confirm-gated fire."""
        s = self._session(req["session_id"])
        ev = req.get("event") or {}
        s.emit("user_event", **ev)
        if ev.get("type") == "confirm_transition":
            s.pending_confirm[ev.get("transition", "")] = True
            await self._fire_ready(s)
        return {"ok": True}

    @rpc()
    async def PopBotEvents(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Long-poll the event ring from an offset (the SSE/websocket chat
        stream analog)."""
        s = self._session(req["session_id"])
        offset = int(req.get("offset", 0))
        timeout = min(float(req.get("timeout", 10.0)), 30.0)
        if offset >= len(s.events):
            s._event_waiter.clear()
            try:
                await asyncio.wait_for(s._event_waiter.wait(), timeout=timeout)
            except asyncio.TimeoutError:
                pass
        return {"events": s.events[offset:], "next_offset": len(s.events)}

    @rpc()
    async def GetBotState(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        s = self._session(req["session_id"])
        return {
            "markers": {loc: list(ms) for loc, ms in s.markers.items()},
            "events": len(s.events),
            "transitions": {
                n: {"satisfied": s.satisfied(t), "stub_id": s.transition_stubs.get(n)}
                for n, t in s.cfg.transitions.items()
            },
        }

    # -- the net engine ------------------------------------------------------
    async def _fire_ready(self, s: BotSession) -> None:
        for name, t in s.cfg.transitions.items():
            while s.satisfied(t):
                if t.confirm and not s.pending_confirm.pop(name, False):
                    s.emit("awaiting_confirm", transition=name)
                    break
                consumed = {
                    loc: [s.markers[loc].pop(0) for _ in range(n)]
                    for loc, n in t.inputs.items()
                }
                s.emit("transition_fired", transition=name, inputs=consumed)
                asyncio.ensure_future(self._run_transition(s, name, t, consumed))

    async def _run_transition(
        self, s: BotSession, name: str, t: BotTransitionConfig, consumed: Dict[str, List[Dict]]
    ) -> None:
        """One firing = one function task through the normal container path."""
        try:
            task = await self.gw.dispatcher.send_and_execute(
                "function",
                getattr(s, "workspace_id", "default"),
                s.transition_stubs[name],
                {"kwargs": {"markers": consumed}},
            )
            done = await self.gw.dispatcher.wait(task.task_id, timeout=t.timeout + 30.0)
            result = await self.gw.dispatcher.get_result(task.task_id)
            if done.status.value != "COMPLETE":
                s.emit("transition_failed", transition=name, status=done.status.value,
                       error=str(result)[:500])
                return
            out_markers = result.get("markers", {}) if isinstance(result, dict) else {}
            pushed = {}
            for loc in t.outputs:
                for m in out_markers.get(loc, []):
                    s.markers.setdefault(loc, []).append(dict(m))
                    pushed.setdefault(loc, []).append(m)
            s.emit("transition_complete", transition=name, outputs=pushed)
            if pushed:
                await self._fire_ready(s)  # cascading fires
        except Exception as e:
            log.exception("bot transition %s failed", name)
            s.emit("transition_failed", transition=name, error=str(e))
