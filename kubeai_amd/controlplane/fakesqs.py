"""In-process fake SQS server (AWS query protocol subset) — the localstack
analog for messenger contract tests: SendMessage / ReceiveMessage with
long polling and visibility timeouts / DeleteMessage /
ChangeMessageVisibility. Queue names come from the URL path's last
segment; queues are created on first use.

Also runnable standalone:  python -m kubeai_amd.controlplane.fakesqs
"""
from __future__ import annotations

import threading
import time
import uuid
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse


class _Queue:
    def __init__(self):
        self.lock = threading.Lock()
        self.cond = threading.Condition(self.lock)
        # message id -> {body, visible_at}
        self.msgs: dict[str, dict] = {}
        # receipt handle -> message id (latest receive wins, AWS-style)
        self.handles: dict[str, str] = {}


class FakeSqsServer:
    def __init__(self, port: int = 0, host: str = "127.0.0.1"):
        queues: dict[str, _Queue] = {}
        qlock = threading.Lock()

        def queue_of(path: str) -> _Queue:
            name = path.rstrip("/").rsplit("/", 1)[-1] or "default"
            with qlock:
                return queues.setdefault(name, _Queue())

        self.queues = queues
        self.queue_of = queue_of

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _send(self, xml: str, code: int = 200):
                body = xml.encode()
                self.send_response(code)
                self.send_header("Content-Type", "text/xml")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_POST(self):
                n = int(self.headers.get("Content-Length") or 0)
                form = parse_qs(self.rfile.read(n).decode())
                action = (form.get("Action") or [""])[0]
                q = queue_of(urlparse(self.path).path)
                if action == "SendMessage":
                    mid = uuid.uuid4().hex
                    with q.cond:
                        q.msgs[mid] = {
                            "body": (form.get("MessageBody") or [""])[0],
                            "visible_at": 0.0,
                        }
                        q.cond.notify_all()
                    return self._send(
                        f"<SendMessageResponse><SendMessageResult>"
                        f"<MessageId>{mid}</MessageId>"
                        f"</SendMessageResult></SendMessageResponse>"
                    )
                if action == "ReceiveMessage":
                    wait = float((form.get("WaitTimeSeconds") or ["0"])[0])
                    vis = float((form.get("VisibilityTimeout") or ["30"])[0])
                    deadline = time.monotonic() + wait
                    while True:
                        with q.cond:
                            now = time.time()
                            got = None
                            for mid, m in q.msgs.items():
                                if m["visible_at"] <= now:
                                    got = (mid, m)
                                    break
                            if got is not None:
                                mid, m = got
                                m["visible_at"] = now + vis
                                rh = uuid.uuid4().hex
                                q.handles[rh] = mid
                                return self._send(
                                    "<ReceiveMessageResponse>"
                                    "<ReceiveMessageResult><Message>"
                                    f"<MessageId>{mid}</MessageId>"
                                    f"<ReceiptHandle>{rh}</ReceiptHandle>"
                                    f"<Body>{m['body']}</Body>"
                                    "</Message></ReceiveMessageResult>"
                                    "</ReceiveMessageResponse>"
                                )
                            if time.monotonic() >= deadline:
                                return self._send(
                                    "<ReceiveMessageResponse>"
                                    "<ReceiveMessageResult/>"
                                    "</ReceiveMessageResponse>"
                                )
                            q.cond.wait(
                                min(0.05, deadline - time.monotonic())
                            )
                if action == "DeleteMessage":
                    rh = (form.get("ReceiptHandle") or [""])[0]
                    with q.cond:
                        mid = q.handles.pop(rh, None)
                        if mid is not None:
                            q.msgs.pop(mid, None)
                    return self._send(
                        "<DeleteMessageResponse/>"
                    )
                if action == "ChangeMessageVisibility":
                    rh = (form.get("ReceiptHandle") or [""])[0]
                    vis = float((form.get("VisibilityTimeout") or ["0"])[0])
                    with q.cond:
                        mid = q.handles.get(rh)
                        if mid is not None and mid in q.msgs:
                            q.msgs[mid]["visible_at"] = time.time() + vis
                            q.cond.notify_all()
                    return self._send(
                        "<ChangeMessageVisibilityResponse/>"
                    )
                return self._send(
                    "<ErrorResponse><Error><Code>InvalidAction</Code>"
                    "</Error></ErrorResponse>", code=400,
                )

        self._httpd = ThreadingHTTPServer((host, port), Handler)
        self._httpd.daemon_threads = True
        self.port = self._httpd.server_address[1]
        self.url = f"http://{host}:{self.port}"
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, daemon=True
        )

    def start(self) -> "FakeSqsServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._httpd.shutdown()
        self._httpd.server_close()


def main():
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=9324)
    args = p.parse_args()
    srv = FakeSqsServer(port=args.port).start()
    print(f"fake sqs on {srv.url}")
    try:
        while True:
            time.sleep(60)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
