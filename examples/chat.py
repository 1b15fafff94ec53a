"""Example chat client against the gateway (reference parity:
examples/chat/chat.py — the official ollama client pointed at :9001; any
Ollama-compatible client works against this gateway)."""

import json
import sys
import urllib.request

GATEWAY = sys.argv[1] if len(sys.argv) > 1 else "http://localhost:9001"
MODEL = sys.argv[2] if len(sys.argv) > 2 else "llama3-8b"

req = urllib.request.Request(
    f"{GATEWAY}/api/chat",
    data=json.dumps({
        "model": MODEL,
        "messages": [{"role": "user", "content": "Why is the sky blue?"}],
    }).encode(),
    headers={"Content-Type": "application/json"})
with urllib.request.urlopen(req, timeout=300) as r:
    resp = json.load(r)
print(f"[{resp.get('worker_id', '?')}] {resp['message']['content']}")

# Streamed variant: stream=true returns NDJSON chunks (one token delta per
# line; final line has done=true). Pass --stream as the third argument.
if "--stream" in sys.argv:
    req = urllib.request.Request(
        f"{GATEWAY}/api/chat",
        data=json.dumps({
            "model": MODEL, "stream": True,
            "messages": [{"role": "user", "content": "Tell me a story."}],
        }).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=300) as r:
        for line in r:
            chunk = json.loads(line)
            print(chunk.get("message", {}).get("content", ""),
                  end="", flush=True)
            if chunk.get("done"):
                print()
