#!/usr/bin/env python3
"""Out-of-process file-parser backend (the second real OoP module; the
reference's analog is the calculator/calculator-gateway pair —
examples/oop-modules — where a child process serves requests through a
directory-resolved endpoint).

Spawned by module-orchestrator (`runtime.type: oop`): reads its config
from MODKIT_MODULE_CONFIG, serves `POST /parse` (raw bytes body,
x-filename / x-markdown headers) on a loopback port, registers that
endpoint with the directory, and heartbeats.  The in-process file-parser
module routes the configured extensions here (config `remote_backends`).

Built-in backend: reStructuredText (.rst) -> text/markdown.
"""
import json
import os
import re
import threading
import time
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

cfg = json.loads(os.environ.get("MODKIT_MODULE_CONFIG", "{}"))
directory = os.environ["MODKIT_DIRECTORY_ENDPOINT"]
name = os.environ.get("MODKIT_MODULE_NAME", "fileparser-oop")


def rst_parse(text: str, markdown: bool) -> str:
    """Tiny RST reader: section underlines -> headings, directives and
    field lists dropped, inline roles unwrapped."""
    out = []
    lines = text.splitlines()
    i = 0
    while i < len(lines):
        line = lines[i]
        nxt = lines[i + 1] if i + 1 < len(lines) else ""
        if (line.strip() and nxt and len(nxt) >= len(line.rstrip())
                and re.fullmatch(r"([=\-~^\"'#*+])\1*", nxt.strip())):
            level = {"=": 1, "-": 2, "~": 3}.get(nxt.strip()[0], 3)
            title = line.strip()
            out.append(("#" * level + " " + title) if markdown else title)
            i += 2
            continue
        if line.lstrip().startswith(".. ") or re.match(r"^:\w+:", line):
            i += 1
            continue
        # inline roles/emphasis: ``code`` stays in md, unwraps in text
        s = re.sub(r":\w+:`([^`]*)`", r"\1", line)
        if not markdown:
            s = s.replace("``", "").replace("**", "").replace("*", "")
        out.append(s)
        i += 1
    return "\n".join(out).strip() + "\n"


class Handler(BaseHTTPRequestHandler):
    def do_POST(self):
        if self.path != "/parse":
            self.send_error(404)
            return
        n = int(self.headers.get("content-length", 0))
        body = self.rfile.read(n) if n else b""
        fname = self.headers.get("x-filename", "doc.rst")
        md = self.headers.get("x-markdown", "0") == "1"
        content = rst_parse(body.decode("utf-8", errors="replace"), md)
        out = json.dumps({
            "filename": fname,
            "backend": "oop-rst",
            "format": "markdown" if md else "text",
            "content": content,
            "greeting": cfg.get("greeting", ""),
        }).encode()
        self.send_response(200)
        self.send_header("content-type", "application/json")
        self.send_header("content-length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)

    def log_message(self, *a):
        pass


def main():
    srv = ThreadingHTTPServer(("127.0.0.1", int(cfg.get("port", 0))),
                              Handler)
    port = srv.server_address[1]
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    endpoint = f"http://127.0.0.1:{port}"
    print(f"oop-fileparser serving {endpoint}", flush=True)
    req = urllib.request.Request(
        directory + "/instances/register", method="POST",
        data=json.dumps({"name": name, "endpoint": endpoint,
                         "meta": {"extensions": cfg.get(
                             "extensions", ["rst"])}}).encode(),
        headers={"content-type": "application/json"})
    iid = json.loads(urllib.request.urlopen(req, timeout=5).read())["id"]
    print("registered as", iid, flush=True)
    while True:
        urllib.request.urlopen(urllib.request.Request(
            directory + f"/instances/{iid}/heartbeat", method="POST",
            data=b""), timeout=5)
        time.sleep(1.0)


if __name__ == "__main__":
    main()
