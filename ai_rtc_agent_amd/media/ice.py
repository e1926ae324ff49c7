"""ICE/TURN server provisioning.

Parity with reference agent.py:80-120 (Twilio ephemeral TURN tokens +
RTCIceServer list + WHIP Link-header builder). Generic REST shape so any
Twilio-compatible token endpoint works:

- TURN_TOKEN_URL + TURN_TOKEN_AUTH: POST, expects {"ice_servers":[...]}
- TWILIO_ACCOUNT_SID / TWILIO_AUTH_TOKEN: the reference's exact env pair,
  hitting the Twilio Tokens endpoint (no twilio SDK needed — plain REST).
"""
from __future__ import annotations

import logging
import os
from dataclasses import dataclass
from typing import List, Optional

import requests

logger = logging.getLogger(__name__)


@dataclass
class IceServer:
    urls: List[str]
    username: Optional[str] = None
    credential: Optional[str] = None


def get_twilio_token() -> Optional[dict]:
    """reference agent.py:80-91 (via REST instead of the twilio SDK)."""
    sid = os.environ.get("TWILIO_ACCOUNT_SID")
    auth = os.environ.get("TWILIO_AUTH_TOKEN")
    if not sid or not auth:
        return None
    try:
        resp = requests.post(
            f"https://api.twilio.com/2010-04-01/Accounts/{sid}/Tokens.json",
            auth=(sid, auth),
            timeout=10,
        )
        resp.raise_for_status()
        return resp.json()
    except requests.RequestException:
        logger.warning("twilio token fetch failed", exc_info=True)
        return None


def get_ice_servers() -> List[IceServer]:
    """reference agent.py:94-109: default STUN + ephemeral TURN if provisioned."""
    servers = [IceServer(urls=["stun:stun.l.google.com:19302"])]
    token = get_twilio_token()
    if token and "ice_servers" in token:
        for s in token["ice_servers"]:
            urls = s.get("urls") or s.get("url")
            if isinstance(urls, str):
                urls = [urls]
            servers.append(
                IceServer(urls=urls, username=s.get("username"), credential=s.get("credential"))
            )
    url = os.environ.get("TURN_TOKEN_URL")
    if url:
        try:
            headers = {}
            if os.environ.get("TURN_TOKEN_AUTH"):
                headers["Authorization"] = os.environ["TURN_TOKEN_AUTH"]
            resp = requests.post(url, headers=headers, timeout=10)
            resp.raise_for_status()
            for s in resp.json().get("ice_servers", []):
                urls = s.get("urls") or [s.get("url")]
                if isinstance(urls, str):
                    urls = [urls]
                servers.append(
                    IceServer(urls=urls, username=s.get("username"), credential=s.get("credential"))
                )
        except requests.RequestException:
            logger.warning("TURN token fetch failed", exc_info=True)
    return servers


def get_link_headers(servers: List[IceServer]) -> List[str]:
    """WHIP Link-header builder (reference agent.py:113-120)."""
    links = []
    for s in servers:
        for url in s.urls:
            link = f'<{url}>; rel="ice-server"'
            if s.username:
                link += f'; username="{s.username}"; credential="{s.credential}"; credential-type="password"'
            links.append(link)
    return links
