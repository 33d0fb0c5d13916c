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
    logger.info(
        json.dumps(
            {
                "service_name": service_name,
                "type": "InferenceData",
                "request_id": request_id,
                "data": data_json,
            }
        )
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
