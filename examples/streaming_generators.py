"""Streaming generators: consume results while the producer still runs.

num_returns="streaming" turns a generator task (or actor method) into an
ObjectRefGenerator: each yielded value is shipped to the caller the
moment it is produced, so a consumer pipeline overlaps with production
— the pattern Serve response streaming and progressive data loading
build on.

    python examples/streaming_generators.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray  # ant_ray_amd alias

ray.init(num_cpus=4)


@ray.remote(num_returns="streaming")
def produce(n):
    for i in range(n):
        time.sleep(0.2)  # pretend each shard takes work
        yield {"shard": i, "rows": 1000 + i}


@ray.remote
class Tokenizer:
    @ray.method(num_returns="streaming")
    def stream_tokens(self, text):
        for word in text.split():
            yield word.upper()


t0 = time.time()
print("== task streaming (consumer overlaps producer)")
for ref in produce.remote(5):
    shard = ray.get(ref)
    print(f"  +{time.time() - t0:4.1f}s got shard {shard['shard']} "
          f"({shard['rows']} rows)")

print("== actor method streaming")
tok = Tokenizer.remote()
print(" ", [ray.get(r) for r in tok.stream_tokens.remote("stream me now")])

print("== ray.cancel on a streaming consumer's sibling")


@ray.remote
def slow_square(x):
    for _ in range(50):
        time.sleep(0.1)
    return x * x


doomed = slow_square.remote(7)
time.sleep(0.3)
ray.cancel(doomed)
try:
    ray.get(doomed, timeout=30)
except Exception as e:
    print(f"  cancelled as expected: {type(e).__name__}")

ray.shutdown()
print("done")
