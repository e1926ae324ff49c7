#!/bin/bash
# parity: reference runpod/start.sh — agent in the background + handler
python /app/agent_main.py --port 8888 --udp-ports 40000-40100 &
python /app/runpod/handler.py
