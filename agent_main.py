#!/usr/bin/env python3
"""Entry point: `python agent_main.py --model-id ... --port 8888`
(parity with `python agent.py ...` in the reference, agent.py:440-474)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ai_rtc_agent_amd.agent import main

if __name__ == "__main__":
    main()
