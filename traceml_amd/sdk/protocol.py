"""File-based RPC protocol between user code and the aggregator
(reference: sdk/protocol.py:143-180).

Rank 0 writes a request file under the session's ``control/`` directory;
the aggregator's FinalSummaryService polls for it, generates the summary
from SQLite, and writes the response artifacts. Paths are the contract.
"""

from __future__ import annotations

import os

CONTROL_DIR = "control"
FINAL_SUMMARY_REQUEST = "final_summary_request.json"
FINAL_SUMMARY_RESPONSE = "final_summary_response.json"

FINAL_SUMMARY_JSON = "final_summary.json"
FINAL_SUMMARY_TXT = "final_summary.txt"
FINAL_SUMMARY_HTML = "final_summary.html"

AGGREGATOR_DB = os.path.join("aggregator", "telemetry.sqlite")


def control_dir(session_dir: str) -> str:
    return os.path.join(session_dir, CONTROL_DIR)


def request_path(session_dir: str) -> str:
    return os.path.join(session_dir, CONTROL_DIR, FINAL_SUMMARY_REQUEST)


def response_path(session_dir: str) -> str:
    return os.path.join(session_dir, CONTROL_DIR, FINAL_SUMMARY_RESPONSE)


def summary_json_path(session_dir: str) -> str:
    return os.path.join(session_dir, FINAL_SUMMARY_JSON)


def summary_txt_path(session_dir: str) -> str:
    return os.path.join(session_dir, FINAL_SUMMARY_TXT)


def summary_html_path(session_dir: str) -> str:
    return os.path.join(session_dir, FINAL_SUMMARY_HTML)


def sqlite_path(session_dir: str) -> str:
    return os.path.join(session_dir, AGGREGATOR_DB)
