"""Runpod serverless glue — parity with reference runpod/handler.py.

Starts nothing itself (start.sh launches the agent in the background);
the handler polls agent health (reference runpod/handler.py:11-27),
publishes connection details as job progress (:41-47) and keeps the job
alive for agent_timeout seconds (default 600, :8,50).
"""
from __future__ import annotations

import os
import time

import requests

AGENT_URL = "http://127.0.0.1:8888"
DEFAULT_TIMEOUT = 600


def wait_for_agent(timeout: int = 120) -> bool:
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if requests.get(AGENT_URL + "/", timeout=2).status_code == 200:
                return True
        except requests.RequestException:
            pass
        time.sleep(2)
    return False


def handler(job):
    job_input = job.get("input", {}) or {}
    agent_timeout = int(job_input.get("agent_timeout", DEFAULT_TIMEOUT))

    if not wait_for_agent():
        return {"error": "agent did not become healthy"}

    pod_id = os.environ.get("RUNPOD_POD_ID", "local")
    public_ip = os.environ.get("RUNPOD_PUBLIC_IP", "127.0.0.1")
    public_port = os.environ.get("RUNPOD_TCP_PORT_8888", "8888")

    yield {
        "status": "ready",
        "pod_id": pod_id,
        "public_ip": public_ip,
        "public_port": public_port,
    }

    # keep-alive window for the WebRTC session (reference :50)
    deadline = time.time() + agent_timeout
    while time.time() < deadline:
        time.sleep(5)
        try:
            requests.get(AGENT_URL + "/", timeout=2)
        except requests.RequestException:
            yield {"status": "agent unhealthy"}
            return
    yield {"status": "timeout reached"}


if __name__ == "__main__":
    try:
        import runpod

        runpod.serverless.start({"handler": handler, "return_aggregate_stream": True})
    except ImportError:
        print("runpod SDK not installed; handler importable for tests")
