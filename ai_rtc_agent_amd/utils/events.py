"""Stream lifecycle webhooks.

Parity with reference lib/events.py:11-63: POSTs StreamStarted/StreamEnded
(stream_id, room_id, timestamp) to WEBHOOK_URL with a Bearer AUTH_TOKEN;
silently disabled when the env vars are unset. Fired from the connection-
state hook (reference agent.py:191-196). Async (thread offload) so the
media loop never blocks on the webhook endpoint.
"""
from __future__ import annotations

import asyncio
import logging
import os
import time
from typing import Optional

import requests
from pydantic import BaseModel

logger = logging.getLogger(__name__)


class WebhookEvent(BaseModel):
    event: str  # StreamStarted | StreamEnded
    stream_id: str
    room_id: Optional[str] = None
    timestamp: int


class StreamEventHandler:
    def __init__(
        self,
        webhook_url: Optional[str] = None,
        auth_token: Optional[str] = None,
    ):
        self.webhook_url = webhook_url or os.environ.get("WEBHOOK_URL")
        self.auth_token = auth_token or os.environ.get("AUTH_TOKEN")

    @property
    def enabled(self) -> bool:
        return bool(self.webhook_url)

    def _post(self, event: WebhookEvent) -> None:
        headers = {"Content-Type": "application/json"}
        if self.auth_token:
            headers["Authorization"] = f"Bearer {self.auth_token}"
        try:
            requests.post(
                self.webhook_url, data=event.model_dump_json(), headers=headers, timeout=5
            )
        except requests.RequestException:
            logger.warning("webhook delivery failed", exc_info=True)

    async def send(self, event_name: str, stream_id: str, room_id: Optional[str] = None) -> None:
        if not self.enabled:
            return
        ev = WebhookEvent(
            event=event_name, stream_id=stream_id, room_id=room_id,
            timestamp=int(time.time() * 1000),
        )
        await asyncio.get_event_loop().run_in_executor(None, self._post, ev)

    async def stream_started(self, stream_id: str, room_id: Optional[str] = None) -> None:
        await self.send("StreamStarted", stream_id, room_id)

    async def stream_ended(self, stream_id: str, room_id: Optional[str] = None) -> None:
        await self.send("StreamEnded", stream_id, room_id)
