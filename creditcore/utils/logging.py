"""Structured JSON request logging.

Keeps the reference's observability contract — two JSON documents per request
on stdout, correlated by request_id (reference app/main.py:57-84):

    {"service_name": ..., "type": "InferenceData", "request_id": ..., "data": ...}
    {"service_name": ..., "type": "ModelOutput",   "request_id": ..., "data": ...}

and extends each ModelOutput line with latency_ms, batch rows and the GPU id
that served the batch (SURVEY.md §5.5 'new framework' note).
"""

from __future__ import annotations

import json
import logging

logger = logging.getLogger("creditcore.requests")


def log_inference_data(service_name: str, request_id: str, data_json: str) -> None:
    """``data_json`` must already be JSON (the raw request body): it is
    embedded verbatim as a nested JSON value instead of being re-escaped
    into a string — re-escaping a ~0.6 MB body cost ~4 ms per request."""
    logger.info(
        '{"service_name": %s, "type": "InferenceData", "request_id": "%s", "data": %s}',
        json.dumps(service_name),
        request_id,
        data_json,
    )


def log_model_output_raw(
    service_name: str,
    request_id: str,
    response_json: str,
    latency_ms: float | None = None,
    rows: int | None = None,
    device: str | None = None,
) -> None:
    """Like log_model_output but embeds an already-serialized response JSON
    verbatim (the serving wire-out fast path)."""
    logger.info(
        '{"service_name": %s, "type": "ModelOutput", "request_id": "%s", '
        '"latency_ms": %.3f, "rows": %s, "device": "%s", "data": %s}',
        json.dumps(service_name),
        request_id,
        latency_ms if latency_ms is not None else -1.0,
        rows if rows is not None else "null",
        device or "",
        response_json,
    )


def log_model_output(
    service_name: str,
    request_id: str,
    output: dict,
    latency_ms: float | None = None,
    rows: int | None = None,
    device: str | None = None,
) -> None:
    doc = {
        "service_name": service_name,
        "type": "ModelOutput",
        "request_id": request_id,
        "data": output,
    }
    if latency_ms is not None:
        doc["latency_ms"] = round(latency_ms, 3)
    if rows is not None:
        doc["rows"] = rows
    if device is not None:
        doc["device"] = device
    logger.info(json.dumps(doc))
