#!/usr/bin/env python3
"""CLI client for the REST text-generation server (reference
tools/text_generation_cli.py).

  python tools/text_generation_cli.py <host:port> "prompt text"
"""
import json
import sys
import urllib.request


def main():
    host = sys.argv[1] if len(sys.argv) > 1 else "127.0.0.1:5000"
    if len(sys.argv) > 2:
        prompts = [sys.argv[2]]
    else:
        print("enter a prompt (ctrl-d to quit):")
        prompts = [sys.stdin.readline().strip()]
    body = json.dumps({"prompts": prompts, "tokens_to_generate": 64,
                       "top_k": 1}).encode()
    req = urllib.request.Request(
        f"http://{host}/api", data=body, method="PUT",
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req) as r:
        out = json.loads(r.read())
    for t in out.get("text", []):
        print(t)


if __name__ == "__main__":
    main()
