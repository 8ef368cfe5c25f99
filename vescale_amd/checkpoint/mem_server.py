"""In-memory checkpoint server — multi-node restore fan-in.

Reference capability: vescale/checkpoint utilities/server/mem_server_lib.py
(a gRPC daemon holding checkpoint bytes in memory so restarted peers can
restore without a shared filesystem).  MI355X-native re-design: a stdlib
ThreadingHTTPServer on the serving rank holds the DCP directory's files in
RAM; peers fetch a manifest and stream the files into a local directory,
then run the normal DCP load.  No gRPC/proto toolchain, no daemon
lifecycle — the server lives inside the training process (or a
`python -m vescale_amd.checkpoint.mem_server <dir>` one-liner) and dies
with it.

Single-node restores don't need this: `mem_checkpoint_path()` (tmpfs) is
zero-copy there.  This server is the cross-host path.

Usage:
    srv = MemCheckpointServer()
    srv.put_dir(ckpt_dir)              # snapshot a saved DCP dir into RAM
    host, port = srv.start()           # begin serving
    # on any peer host:
    fetch_checkpoint(f"http://{host}:{port}", local_dir)
    checkpoint.load(local_dir, state)  # ordinary DCP load
"""
from __future__ import annotations

import json
import os
import threading
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Dict, Optional, Tuple

__all__ = ["MemCheckpointServer", "fetch_checkpoint"]


class MemCheckpointServer:
    def __init__(self):
        self._files: Dict[str, bytes] = {}
        self._httpd: Optional[ThreadingHTTPServer] = None
        self._thread: Optional[threading.Thread] = None

    # ---------------- content management ----------------
    def put_dir(self, path: str) -> int:
        """Snapshot every file under `path` (recursively) into memory;
        returns total bytes held."""
        total = 0
        for root, _dirs, files in os.walk(path):
            for f in files:
                full = os.path.join(root, f)
                rel = os.path.relpath(full, path)
                with open(full, "rb") as fh:
                    data = fh.read()
                self._files[rel] = data
                total += len(data)
        return total

    def put_file(self, rel: str, data: bytes) -> None:
        self._files[rel] = data

    def clear(self) -> None:
        self._files.clear()

    # ---------------- serving ----------------
    def start(self, host: str = "0.0.0.0", port: int = 0) -> Tuple[str, int]:
        files = self._files

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # quiet
                pass

            def do_GET(self):
                rel = self.path.lstrip("/")
                if rel == "__manifest__":
                    body = json.dumps(
                        {"files": {k: len(v) for k, v in files.items()}}
                    ).encode()
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    return
                data = files.get(rel)
                if data is None:
                    self.send_response(404)
                    self.end_headers()
                    return
                self.send_response(200)
                self.send_header("Content-Type", "application/octet-stream")
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

        self._httpd = ThreadingHTTPServer((host, port), Handler)
        self._thread = threading.Thread(target=self._httpd.serve_forever, daemon=True)
        self._thread.start()
        h, p = self._httpd.server_address[:2]
        return (host if host != "0.0.0.0" else "127.0.0.1", p)

    def stop(self) -> None:
        if self._httpd is not None:
            self._httpd.shutdown()
            self._httpd.server_close()
            self._httpd = None
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None


def fetch_checkpoint(url: str, dest_dir: str, timeout: float = 60.0) -> int:
    """Download a served checkpoint into `dest_dir`; returns bytes read.
    The result is a normal on-disk DCP directory for checkpoint.load."""
    with urllib.request.urlopen(f"{url}/__manifest__", timeout=timeout) as r:
        manifest = json.loads(r.read())
    total = 0
    for rel, size in manifest["files"].items():
        out = os.path.normpath(os.path.join(dest_dir, rel))
        if not out.startswith(os.path.normpath(dest_dir) + os.sep):
            raise ValueError(f"refusing path outside dest_dir: {rel!r}")
        os.makedirs(os.path.dirname(out) or dest_dir, exist_ok=True)
        with urllib.request.urlopen(f"{url}/{rel}", timeout=timeout) as r:
            data = r.read()
        assert len(data) == size, f"{rel}: got {len(data)} want {size}"
        with open(out, "wb") as fh:
            fh.write(data)
        total += len(data)
    return total


def _main():
    import argparse

    ap = argparse.ArgumentParser(description="serve a DCP checkpoint dir from RAM")
    ap.add_argument("dir")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=0)
    args = ap.parse_args()
    srv = MemCheckpointServer()
    n = srv.put_dir(args.dir)
    host, port = srv.start(args.host, args.port)
    print(f"serving {n} bytes at http://{host}:{port}", flush=True)
    threading.Event().wait()


if __name__ == "__main__":
    _main()
