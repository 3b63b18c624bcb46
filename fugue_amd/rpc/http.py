"""HTTP RPC server/client for distributed worker→driver callbacks.

Reference parity: ``fugue/rpc/flask.py`` (FlaskRPCServer: HTTP POST
/invoke with json-encoded args).  Implementation uses the stdlib
``http.server`` (no flask dependency).  Conf keys::

    fugue.rpc.server: fugue_amd.rpc.http.HttpRPCServer
    fugue.rpc.http.host / fugue.rpc.http.port
"""
import base64
import json
import pickle
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any
from urllib import request as urlrequest

from fugue_amd.rpc.base import RPCClient, RPCServer


class HttpRPCClient(RPCClient):
    def __init__(self, url: str, key: str):
        self._url = url
        self._key = key

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        payload = json.dumps(
            dict(
                key=self._key,
                args=base64.b64encode(pickle.dumps((args, kwargs))).decode(),
            )
        ).encode()
        req = urlrequest.Request(
            self._url + "/invoke",
            data=payload,
            headers={"Content-Type": "application/json"},
        )
        with urlrequest.urlopen(req, timeout=60) as resp:
            body = json.loads(resp.read().decode())
        if body.get("error"):
            raise RuntimeError(f"rpc error: {body['error']}")
        return pickle.loads(base64.b64decode(body["result"]))


class HttpRPCServer(RPCServer):
    """Reference parity: ``fugue/rpc/flask.py:17``."""

    def __init__(self, conf: Any):
        super().__init__(conf)
        self._host = self.conf.get("fugue.rpc.http.host", "127.0.0.1")
        self._port = int(self.conf.get("fugue.rpc.http.port", 0))
        self._httpd: Any = None
        self._thread: Any = None

    def make_client(self, handler: Any) -> RPCClient:
        key = self.register(handler)
        host, port = self._httpd.server_address[:2]
        return HttpRPCClient(f"http://{host}:{port}", key)

    def start_server(self) -> None:
        server = self

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self) -> None:  # noqa: N802
                try:
                    length = int(self.headers.get("Content-Length", "0"))
                    body = json.loads(self.rfile.read(length).decode())
                    args, kwargs = pickle.loads(
                        base64.b64decode(body["args"])
                    )
                    result = server.invoke(body["key"], *args, **kwargs)
                    out = json.dumps(
                        dict(
                            result=base64.b64encode(
                                pickle.dumps(result)
                            ).decode()
                        )
                    ).encode()
                    self.send_response(200)
                except Exception as e:  # pragma: no cover
                    out = json.dumps(dict(error=str(e))).encode()
                    self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(out)))
                self.end_headers()
                self.wfile.write(out)

            def log_message(self, *a: Any) -> None:  # silence
                ...

        self._httpd = ThreadingHTTPServer((self._host, self._port), Handler)
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, daemon=True
        )
        self._thread.start()

    def stop_server(self) -> None:
        if self._httpd is not None:
            self._httpd.shutdown()
            self._httpd.server_close()
            self._httpd = None
