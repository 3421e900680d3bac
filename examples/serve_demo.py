"""Ray Serve: composed app behind the HTTP proxy.

    python examples/serve_demo.py
    curl -X POST localhost:8000/summarize -d 'hello world'
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray
from ray import serve


@serve.deployment(num_replicas=2)
class Tokenizer:
    def __call__(self, text: str):
        return text.split()


@serve.deployment
class Summarizer:
    def __init__(self, tokenizer):
        self.tokenizer = tokenizer

    async def __call__(self, request):
        text = (await request.body()).decode()
        tokens = await self.tokenizer.remote(text)
        return {"n_tokens": len(tokens), "first": tokens[:3]}


if __name__ == "__main__":
    ray.init()
    serve.run(Summarizer.bind(Tokenizer.bind()), name="demo",
              route_prefix="/summarize")
    import sys
    import urllib.request

    req = urllib.request.Request("http://127.0.0.1:8000/summarize",
                                 data=b"hello ray serve demo", method="POST")
    with urllib.request.urlopen(req, timeout=30) as r:
        print("self-test:", r.read().decode())
    if "--serve" in sys.argv:
        print("serving on :8000/summarize — ctrl-c to exit")
        import time

        while True:
            time.sleep(5)
    serve.shutdown()
    ray.shutdown()
