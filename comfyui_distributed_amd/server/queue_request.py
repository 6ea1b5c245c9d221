"""/distributed/queue payload schema (parity: reference
api/queue_request.py:5-79)."""

from __future__ import annotations

from dataclasses import dataclass, field


class QueueRequestError(ValueError):
    pass


@dataclass
class QueueRequestPayload:
    prompt: dict
    client_id: str
    enabled_worker_ids: list[str] = field(default_factory=list)
    delegate_master: bool = False
    trace_execution_id: str | None = None
    workflow: dict | None = None
    #: wire-compat field; orchestration always runs with auto-prepare
    #: semantics (reference api/queue_request.py:12-24)
    auto_prepare: bool = True


def parse_queue_request_payload(data: dict) -> QueueRequestPayload:
    if not isinstance(data, dict):
        raise QueueRequestError("payload must be a JSON object")
    prompt = data.get("prompt")
    workflow = data.get("workflow")
    if prompt is None and isinstance(workflow, dict):
        prompt = workflow.get("prompt")
    if not isinstance(prompt, dict) or not prompt:
        raise QueueRequestError("missing or empty 'prompt'")
    enabled = data.get("enabled_worker_ids")
    if enabled is None:
        enabled = data.get("workers")  # alias
    if enabled is None or not isinstance(enabled, list):
        raise QueueRequestError("missing 'enabled_worker_ids'")
    client_id = data.get("client_id")
    if not client_id:
        raise QueueRequestError("missing 'client_id'")
    auto_prepare = data.get("auto_prepare", True)
    if not isinstance(auto_prepare, bool):
        raise QueueRequestError("auto_prepare must be a boolean when provided")
    return QueueRequestPayload(
        auto_prepare=auto_prepare,
        prompt=prompt,
        client_id=str(client_id),
        enabled_worker_ids=[str(w) for w in enabled],
        delegate_master=bool(data.get("delegate_master", False)),
        trace_execution_id=data.get("trace_execution_id"),
        workflow=workflow if isinstance(workflow, dict) else None,
    )
