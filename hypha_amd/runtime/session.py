"""Job-bridge client: HTTP over the worker's Unix socket.

API parity with the reference executor session
(/root/reference/executors/accelerate/src/hypha/accelerate_executor/api.py):
fetch, send_resource, send_status and the SSE receive iterator.
"""

from __future__ import annotations

import json
from typing import Iterator

import httpx


class Session:
    def __init__(self, socket_path: str):
        transport = httpx.HTTPTransport(uds=socket_path)
        self.client = httpx.Client(transport=transport, base_url="http://bridge",
                                   timeout=httpx.Timeout(600.0))

    def fetch(self, reference: dict) -> dict:
        r = self.client.post("/resources/fetch", json=reference)
        r.raise_for_status()
        return r.json()

    def send_resource(self, reference: dict, rel_path: str) -> dict:
        r = self.client.post(
            "/resources/send", json={"reference": reference, "path": rel_path}
        )
        r.raise_for_status()
        return r.json()

    def send_status(self, status: dict) -> dict:
        r = self.client.post("/status/send", json=status)
        r.raise_for_status()
        return r.json()

    def receive(self) -> Iterator[dict]:
        """SSE iterator over received-resource pointer events."""
        with self.client.stream("GET", "/resources/receive") as resp:
            for line in resp.iter_lines():
                if line.startswith("data: "):
                    yield json.loads(line[6:])

    def close(self):
        self.client.close()
