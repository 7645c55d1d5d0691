"""OpenAI-compatible API smoke test against a live serving instance
(equivalent of the reference's examples/vllm/test_openai_api.py)."""
import json
import sys

import requests

BASE = sys.argv[1] if len(sys.argv) > 1 else "http://127.0.0.1:8080/serve/openai"
MODEL = sys.argv[2] if len(sys.argv) > 2 else "test_llm"


def main():
    r = requests.post(BASE + "/v1/chat/completions", json={
        "model": MODEL, "max_tokens": 16, "temperature": 0.8,
        "messages": [{"role": "user", "content": "Hello!"}],
    })
    r.raise_for_status()
    print("chat:", json.dumps(r.json(), indent=1)[:400])

    r = requests.post(BASE + "/v1/completions", json={
        "model": MODEL, "prompt": "The capital of France is",
        "max_tokens": 8, "temperature": 0.0,
    })
    r.raise_for_status()
    print("completion:", r.json()["choices"][0]["text"])

    r = requests.get(BASE + "/v1/models")
    print("models:", [m["id"] for m in r.json()["data"]])


if __name__ == "__main__":
    main()
