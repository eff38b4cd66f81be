#!/usr/bin/env python3
"""Demo OoP module child (reference examples/oop-modules/calculator):
reads MODKIT_MODULE_CONFIG, registers with the directory, heartbeats."""
import json
import os
import time
import urllib.request

cfg = json.loads(os.environ.get("MODKIT_MODULE_CONFIG", "{}"))
base = os.environ["MODKIT_DIRECTORY_ENDPOINT"]
name = os.environ.get("MODKIT_MODULE_NAME", "demo-oop")
print("child config:", cfg, flush=True)
req = urllib.request.Request(
    base + "/instances/register", method="POST",
    data=json.dumps({"name": name, "endpoint": "uds:///tmp/demo-oop.sock",
                     "meta": cfg}).encode(),
    headers={"content-type": "application/json"})
iid = json.loads(urllib.request.urlopen(req, timeout=5).read())["id"]
print("registered as", iid, flush=True)
while True:
    urllib.request.urlopen(urllib.request.Request(
        base + f"/instances/{iid}/heartbeat", method="POST", data=b""),
        timeout=5)
    time.sleep(1.0)
