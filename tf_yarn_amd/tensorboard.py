"""TensorBoard side-task runner.

The environment has no tensorboard package, so the framework serves its own
metrics dashboard: a small HTTP server over the JSONL event files the
framework's SummaryWriter produces (scalars as JSON + a minimal HTML view).
API parity with the reference (``tf_yarn/tensorboard.py``): ``start_tf_board``
reserves a port, runs the server, advertises the URL through the KV ``url``
event; termination timeout via ``TB_TERMINATION_TIMEOUT_SECONDS``.
"""

from __future__ import annotations

import json
import logging
import os
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional, Tuple

from tf_yarn_amd import event
from tf_yarn_amd.kv import KVClient
from tf_yarn_amd.utils import tb

logger = logging.getLogger(__name__)

DEFAULT_TERMINATION_TIMEOUT_SECONDS = 30


def get_termination_timeout() -> float:
    """Reference ``tensorboard.py:19-25``."""
    raw = os.environ.get("TB_TERMINATION_TIMEOUT_SECONDS", "")
    try:
        timeout = float(raw)
    except ValueError:
        timeout = -1
    if timeout < 0:
        timeout = DEFAULT_TERMINATION_TIMEOUT_SECONDS
    return timeout


def url_event_name(task: str) -> str:
    """Reference ``tensorboard.py:52-55``."""
    return f"{task}/url"


def _make_handler(model_dir: str):
    class Handler(BaseHTTPRequestHandler):
        def log_message(self, fmt, *args):  # quiet
            pass

        def do_GET(self):
            if self.path.startswith("/metrics"):
                events = tb.read_events(model_dir)
                body = json.dumps(events).encode()
                ctype = "application/json"
            else:
                events = tb.read_events(model_dir)
                tags = {}
                for e in events:
                    tags.setdefault(e.get("tag", "?"), []).append(
                        (e.get("step"), e.get("value")))
                rows = "".join(
                    f"<tr><td>{t}</td><td>{len(v)}</td>"
                    f"<td>{v[-1][1] if v else ''}</td></tr>"
                    for t, v in sorted(tags.items()))
                body = (f"<html><body><h1>miyarn board</h1>"
                        f"<p>model_dir: {model_dir}</p>"
                        f"<table border=1><tr><th>tag</th><th>points</th>"
                        f"<th>last</th></tr>{rows}</table>"
                        f"<p><a href='/metrics'>raw JSON</a></p>"
                        f"</body></html>").encode()
                ctype = "text/html"
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    return Handler


def start_tf_board(client: Optional[KVClient], model_dir: str,
                   task: Optional[str] = None
                   ) -> Tuple[ThreadingHTTPServer, str]:
    """Start the board server and advertise its URL
    (reference ``tensorboard.py:28-49``)."""
    server = ThreadingHTTPServer(("127.0.0.1", 0),
                                 _make_handler(model_dir))
    thread = threading.Thread(target=server.serve_forever,
                              name="miyarn-board", daemon=True)
    thread.start()
    url = f"http://127.0.0.1:{server.server_port}"
    logger.info("board serving %s at %s", model_dir, url)
    if client is not None and task is not None:
        event.url_event(client, task, url)
    return server, url
