"""GPU e2e: the WHOLE BASELINE config-2 path on an MI355X — C++
hyperspot-server -> UDS engine worker -> gfx950 kernels -> REST/SSE."""

import json
import time
import urllib.request
from pathlib import Path

import pytest

from tests.test_host_e2e import ServerProc, _free_port, _http, BASE

ROOT = Path(__file__).resolve().parent.parent


@pytest.mark.gpu
def test_rest_chat_on_gpu(tmp_path):
    import torch
    assert torch.cuda.is_available()
    port = _free_port()
    sock = str(tmp_path / "llm.sock")
    cfg = f"""
server:
  home_dir: "{tmp_path}"
logging:
  default:
    console_level: warn
modules:
  api-gateway:
    config:
      bind_addr: "127.0.0.1:{port}"
      auth_disabled: true
  llm-gateway:
    config:
      model: "llama3-8b"
      worker_socket: "{sock}"
      auto_start_worker: true
      worker:
        eager: true
        max_num_seqs: 8
        num_gpu_blocks: 512
"""
    cfg_path = tmp_path / "gpu.yaml"
    cfg_path.write_text(cfg)
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        srv.wait_worker(timeout=300)
        url = BASE.format(port)
        # sync completion
        t0 = time.time()
        st, body = _http("POST", url + "/v1/chat/completions",
                         {"model": "llama3-8b",
                          "messages": [{"role": "user", "content":
                                        [{"type": "text",
                                          "text": "hello mi355x"}]}],
                          "max_tokens": 16}, timeout=120)
        assert st == 200, body
        d = json.loads(body)
        assert d["usage"]["output_tokens"] == 16
        assert d["model_used"].endswith("llama3-8b")
        # SSE stream with [DONE]
        req = urllib.request.Request(
            url + "/v1/chat/completions", method="POST",
            data=json.dumps({"model": "llama3-8b", "stream": True,
                             "messages": [{"role": "user", "content":
                                           [{"type": "text",
                                             "text": "stream"}]}],
                             "max_tokens": 8}).encode(),
            headers={"content-type": "application/json"})
        lines = []
        with urllib.request.urlopen(req, timeout=120) as r:
            for raw in r:
                line = raw.decode().strip()
                if line:
                    lines.append(line)
        assert lines[-1] == "data: [DONE]", lines[-3:]
        deltas = [json.loads(x[6:]) for x in lines[:-1]
                  if x.startswith("data: ")]
        assert any("content" in c.get("delta", {}) for c in deltas)
        assert any(c.get("finish_reason") for c in deltas)
        # engine metrics visible through the gateway
        st, body = _http("GET", url + "/metrics")
        assert "hyperspot_kv_occupancy" in body
        print(f"gpu REST e2e ok in {time.time() - t0:.1f}s")
    finally:
        srv.stop()
