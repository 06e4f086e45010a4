#!/usr/bin/env python3
"""OpenAI-compatible API client demo (role of reference examples/chat-api-client.js).

Start the server first:
  ./dllama-api --model <m> --tokenizer <t> --port 9990
"""
import json
import http.client
import sys

host, port = (sys.argv[1] if len(sys.argv) > 1 else "127.0.0.1:9990").split(":")
conn = http.client.HTTPConnection(host, int(port), timeout=300)

# non-streaming
body = json.dumps({"model": "dllama",
                   "messages": [{"role": "user", "content": "What is 2+2?"}],
                   "max_tokens": 64, "temperature": 0.0})
conn.request("POST", "/v1/chat/completions", body,
             {"Content-Type": "application/json"})
resp = json.loads(conn.getresponse().read())
print("assistant:", resp["choices"][0]["message"]["content"])
print("usage:", resp["usage"])

# streaming (SSE)
body = json.dumps({"model": "dllama",
                   "messages": [{"role": "user", "content": "Count to five."}],
                   "max_tokens": 64, "stream": True})
conn.request("POST", "/v1/chat/completions", body,
             {"Content-Type": "application/json"})
r = conn.getresponse()
buf = b""
while True:
    chunk = r.read(1)
    if not chunk:
        break
    buf += chunk
    while b"\n\n" in buf:
        event, buf = buf.split(b"\n\n", 1)
        if not event.startswith(b"data: "):
            continue
        data = event[6:]
        if data == b"[DONE]":
            print()
            sys.exit(0)
        delta = json.loads(data)["choices"][0]["delta"].get("content", "")
        print(delta, end="", flush=True)
